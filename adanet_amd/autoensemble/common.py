"""AutoEnsemble internals: wrap arbitrary estimators/modules as Builders.

Reference: adanet/autoensemble/common.py:28-268 (_BuilderFromSubestimator,
_GeneratorFromCandidatePool, AutoEnsembleSubestimator).
"""

from __future__ import annotations

import dataclasses
from typing import Any, Callable, Dict, List, Optional, Sequence, Union

import torch
from torch import nn

from adanet_amd.subnetwork.generator import (Builder, Generator, Subnetwork,
                                             SubnetworkModule)


@dataclasses.dataclass(frozen=True)
class AutoEnsembleSubestimator:
    """(estimator, train_input_fn, prediction_only) — a per-candidate private
    input pipeline enables bagging (reference common.py:59-93)."""

    estimator: Any
    train_input_fn: Optional[Callable] = None
    prediction_only: bool = False


class _CallableModule(SubnetworkModule):
    """Wraps a user nn.Module whose forward returns logits (or
    (last_layer, logits), or a predictions dict) into the Subnetwork
    contract. ``logits_fn`` / ``last_layer_fn`` extract the respective
    tensors from the raw output (reference common.py:31-40 default digs
    predictions["logits"]; custom last_layer_fn per estimator_test.py:415).
    """

    def __init__(self, module: nn.Module, logits_fn=None, last_layer_fn=None):
        super().__init__()
        self.inner = module
        self._logits_fn = logits_fn
        self._last_layer_fn = last_layer_fn
        d = getattr(module, "last_layer_dim", None)
        if d is not None:
            self.last_layer_dim = d

    def forward(self, features):
        out = self.inner(features)
        raw = out
        if self._logits_fn is not None:
            out = self._logits_fn(out)
        if isinstance(out, tuple):
            last, logits = out
        elif isinstance(out, dict):
            logits = out.get("logits", None)
            if logits is None:
                raise ValueError(
                    "Subestimator predictions need a 'logits' key or a "
                    "custom logits_fn")
            last = out.get("last_layer", logits)
        else:
            last, logits = out, out
        if self._last_layer_fn is not None:
            last = self._last_layer_fn(raw)
        return last, logits


class _BuilderFromSubestimator(Builder):
    """Adapts an estimator-like object into a Builder
    (reference common.py:96-198; complexity = const 0, :188)."""

    def __init__(self, name: str, subestimator: AutoEnsembleSubestimator,
                 logits_fn=None, last_layer_fn=None):
        self._name = name
        self._subestimator = subestimator
        self._logits_fn = logits_fn
        self._last_layer_fn = last_layer_fn

    @property
    def name(self) -> str:
        return self._name

    @property
    def subestimator(self) -> AutoEnsembleSubestimator:
        return self._subestimator

    @property
    def train_input_fn(self):
        return self._subestimator.train_input_fn

    @property
    def prediction_only(self) -> bool:
        return self._subestimator.prediction_only

    def _build_module(self, features, logits_dimension, training):
        est = self._subestimator.estimator
        if hasattr(est, "build_model"):
            return est.build_model(features, logits_dimension, training)
        if isinstance(est, nn.Module):
            return est
        if callable(est):
            return est(features, logits_dimension)
        raise ValueError(
            "candidate_pool entries must expose build_model(), be an "
            "nn.Module, or be callable(features, logits_dimension); got %r" %
            (est,))

    def build_subnetwork(self, features, logits_dimension, training,
                         previous_ensemble=None) -> Subnetwork:
        module = self._build_module(features, logits_dimension, training)
        if not isinstance(module, SubnetworkModule) or (
                self._logits_fn is not None
                or self._last_layer_fn is not None):
            module = _CallableModule(module, self._logits_fn,
                                     self._last_layer_fn)
        return Subnetwork(module=module, complexity=0.0, name=self._name)

    def build_optimizer(self, params, iteration: int = 0):
        if self._subestimator.prediction_only:
            return None
        est = self._subestimator.estimator
        if hasattr(est, "make_optimizer"):
            return est.make_optimizer(params)
        from adanet_amd.ops.optim import FusedSGD
        return FusedSGD(params, lr=0.01)


class _GeneratorFromCandidatePool(Generator):
    """Reference common.py:218-268 (dict pools sorted by name, :235)."""

    def __init__(self, candidate_pool, logits_fn=None, last_layer_fn=None):
        self._pool = candidate_pool
        self._logits_fn = logits_fn
        self._last_layer_fn = last_layer_fn

    def generate_candidates(self, previous_ensemble, iteration_number,
                            previous_ensemble_reports, all_reports,
                            config=None) -> List[Builder]:
        pool = self._pool
        if callable(pool) and not isinstance(pool, (dict, list, tuple)):
            try:
                pool = pool(config, iteration_number)
            except TypeError:
                pool = pool(config)
        builders = []
        if isinstance(pool, dict):
            for name in sorted(pool):  # determinism (reference :235)
                builders.append(
                    self._wrap(name, pool[name]))
        else:
            for i, cand in enumerate(pool):
                name = getattr(cand, "name", None) or "{}{}".format(
                    type(getattr(cand, "estimator", cand)).__name__, i)
                builders.append(self._wrap(name, cand))
        return builders

    def _wrap(self, name, cand) -> Builder:
        if not isinstance(cand, AutoEnsembleSubestimator):
            cand = AutoEnsembleSubestimator(cand)
        return _BuilderFromSubestimator(name, cand,
                                        logits_fn=self._logits_fn,
                                        last_layer_fn=self._last_layer_fn)
