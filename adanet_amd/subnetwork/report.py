"""Reports: structured info a Builder passes to future Generators.

Mirrors reference adanet/subnetwork/report.py:29-196 including its strict
type validation tables (hparams values must be bool/int/float/str; attributes
and metrics may additionally be tensors, materialized to python scalars).
"""

from __future__ import annotations

import dataclasses
from typing import Any, Dict, Optional

import torch

_SCALAR_TYPES = (bool, int, float, str)


def _validate_scalar(name: str, key: str, value: Any, allow_tensor: bool):
    if isinstance(value, torch.Tensor):
        if allow_tensor and value.numel() == 1:
            return
        raise ValueError(
            "%s %r=%r is a non-scalar tensor" % (name, key, value)
            if allow_tensor else
            "%s %r=%r must not be a Tensor" % (name, key, value))
    if not isinstance(value, _SCALAR_TYPES):
        raise ValueError("%s %r has invalid type %s" % (name, key, type(value)))


class Report(object):
    """A container for data a Builder wants to expose to future Generators.

    Reference: adanet/subnetwork/report.py:29-133. ``hparams`` must be python
    scalars; ``attributes`` and ``metrics`` may be python scalars or
    0-d/1-element tensors (the reference allows rank-0 Tensors there).
    """

    def __init__(self, hparams: Dict[str, Any], attributes: Dict[str, Any],
                 metrics: Dict[str, Any]):
        for k, v in dict(hparams).items():
            _validate_scalar("hparam", k, v, allow_tensor=False)
        for k, v in dict(attributes).items():
            _validate_scalar("attribute", k, v, allow_tensor=True)
        for k, v in dict(metrics).items():
            _validate_scalar("metric", k, v, allow_tensor=True)
        self._hparams = dict(hparams)
        self._attributes = dict(attributes)
        self._metrics = dict(metrics)

    @property
    def hparams(self):
        return self._hparams

    @property
    def attributes(self):
        return self._attributes

    @property
    def metrics(self):
        return self._metrics

    def materialize(self, iteration_number: int, name: str,
                    included_in_final_ensemble: bool = False):
        """Evaluates any tensor values into python scalars."""

        def _mat(d):
            out = {}
            for k, v in d.items():
                if isinstance(v, torch.Tensor):
                    v = v.detach().reshape(()).item()
                out[k] = v
            return out

        return MaterializedReport(
            iteration_number=iteration_number,
            name=name,
            hparams=dict(self._hparams),
            attributes=_mat(self._attributes),
            metrics=_mat(self._metrics),
            included_in_final_ensemble=included_in_final_ensemble,
        )


@dataclasses.dataclass(frozen=True)
class MaterializedReport:
    """Materialized (all-python) report.

    Reference: adanet/subnetwork/report.py:136-196.
    """

    iteration_number: int
    name: str
    hparams: Dict[str, Any] = dataclasses.field(default_factory=dict)
    attributes: Dict[str, Any] = dataclasses.field(default_factory=dict)
    metrics: Dict[str, Any] = dataclasses.field(default_factory=dict)
    included_in_final_ensemble: bool = False

    def to_json(self) -> Dict[str, Any]:
        return dataclasses.asdict(self)

    @staticmethod
    def from_json(d: Dict[str, Any]) -> "MaterializedReport":
        return MaterializedReport(**d)
