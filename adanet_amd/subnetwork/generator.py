"""Search-space API: Subnetwork, Builder, Generator.

MI355X-native re-design of the reference search-space contract
(reference: adanet/subnetwork/generator.py:39-330). The reference is a TF1
graph-mode API where ``build_subnetwork`` returns symbolic tensors and
``build_subnetwork_train_op`` returns a graph op. Here the contract is
define-by-run PyTorch: ``build_subnetwork`` returns a :class:`Subnetwork`
wrapping an ``nn.Module`` whose ``forward`` produces ``(last_layer, logits)``,
and ``build_optimizer`` returns an optimizer over the subnetwork's
parameters (the analog of ``TrainOpSpec``, reference
adanet/subnetwork/generator.py:39-58).
"""

from __future__ import annotations

import abc
import dataclasses
from typing import Any, Callable, Dict, List, Optional, Sequence, Tuple

import torch
from torch import nn


class SubnetworkModule(nn.Module):
    """Base class for subnetwork modules.

    ``forward(features) -> (last_layer, logits)``. Subclasses may instead
    override ``forward`` to return just ``logits`` in which case
    ``last_layer`` is taken to equal ``logits``.
    """

    def forward(self, features):  # pragma: no cover - interface
        raise NotImplementedError


@dataclasses.dataclass
class Subnetwork:
    """An ensemble-candidate building block h(x).

    Mirrors the reference namedtuple (reference: adanet/subnetwork/
    generator.py:62-158) with fields re-interpreted for define-by-run:

    - ``module``: an ``nn.Module`` computing ``(last_layer, logits)`` from the
      features. Replaces the reference's symbolic ``last_layer``/``logits``
      tensor fields.
    - ``complexity``: scalar measure r(h) of the subnetwork's complexity.
    - ``shared``: arbitrary cross-iteration state (reference ``shared`` field,
      generator.py:120-138), visible to the next iteration's Generator via
      ``previous_ensemble``.
    - ``name``: set by the framework from the Builder's name.
    """

    module: nn.Module
    complexity: float = 1.0
    shared: Any = None
    name: str = ""

    def __post_init__(self):
        if not isinstance(self.module, nn.Module):
            raise ValueError(
                "Subnetwork.module must be an nn.Module, got %r" % (self.module,))
        c = self.complexity
        if isinstance(c, torch.Tensor):
            c = float(c.detach().cpu())
            self.complexity = c
        if not isinstance(self.complexity, (int, float)):
            raise ValueError("complexity must be a scalar, got %r" % (c,))

    def __call__(self, features):
        out = self.module(features)
        if isinstance(out, tuple):
            last_layer, logits = out
        else:
            last_layer, logits = out, out
        return last_layer, logits


class Builder(abc.ABC):
    """Builds one candidate subnetwork per iteration.

    Mirrors reference adanet/subnetwork/generator.py:162-270. Differences:

    - ``build_subnetwork`` receives concrete example ``features`` (a tensor or
      dict of tensors from the first batch) rather than graph placeholders,
      plus the target device/dtype, and returns a :class:`Subnetwork`.
    - ``build_optimizer(params)`` replaces ``build_subnetwork_train_op``:
      returns a ``torch.optim.Optimizer``-compatible object (our fused HIP
      optimizers in :mod:`adanet_amd.ops.optim` satisfy this).
    - ``build_mixture_weights_optimizer`` (optional) replaces the deprecated
      ``build_mixture_weights_train_op`` (reference generator.py:231-257).
    """

    @property
    @abc.abstractmethod
    def name(self) -> str:
        """Unique name of this subnetwork within an iteration."""

    @abc.abstractmethod
    def build_subnetwork(
        self,
        features,
        logits_dimension: int,
        training: bool,
        previous_ensemble=None,
    ) -> Subnetwork:
        """Builds and returns a Subnetwork for this iteration."""

    def build_optimizer(self, params, iteration: int = 0):
        """Returns an optimizer over ``params``. Default: fused SGD(lr=.01).

        Analog of reference ``build_subnetwork_train_op``
        (adanet/subnetwork/generator.py:198-229).
        """
        from adanet_amd.ops.optim import FusedSGD
        return FusedSGD(params, lr=0.01)

    def build_mixture_weights_optimizer(self, params, iteration: int = 0):
        """Optimizer for this candidate's mixture weights; None = ensembler default."""
        return None

    def build_subnetwork_report(self):
        """Returns a `Report` for Generators at future iterations.

        Reference: adanet/subnetwork/generator.py:259-270.
        """
        return None

    # Back-compat hook honored by the ensemble builder, mirroring reference
    # ensemble_builder.py:371-395.
    def prune_previous_ensemble(self, previous_ensemble) -> List[int]:
        """Returns indices of previous-ensemble subnetworks to keep."""
        return list(range(len(previous_ensemble.subnetworks))) if (
            previous_ensemble is not None) else []


class Generator(abc.ABC):
    """Generates the pool of candidate Builders for an iteration.

    Mirrors reference adanet/subnetwork/generator.py:274-320. MUST be
    deterministic given identical inputs (the distributed round-robin
    placement relies on all ranks generating the same pool in the same
    order — reference generator.py:288 has the same requirement for
    different workers).
    """

    @abc.abstractmethod
    def generate_candidates(
        self,
        previous_ensemble,
        iteration_number: int,
        previous_ensemble_reports: Sequence,
        all_reports: Sequence,
        config=None,
    ) -> List[Builder]:
        """Generates Builders to train this iteration."""


class SimpleGenerator(Generator):
    """Generator that always returns the same list of Builders.

    Reference: adanet/subnetwork/generator.py:323-330.
    """

    def __init__(self, subnetwork_builders: Sequence[Builder]):
        if not subnetwork_builders:
            raise ValueError("subnetwork_builders must not be empty")
        for b in subnetwork_builders:
            if not isinstance(b, Builder):
                raise ValueError("%r is not a Builder" % (b,))
        self._builders = list(subnetwork_builders)

    def generate_candidates(self, previous_ensemble, iteration_number,
                            previous_ensemble_reports, all_reports,
                            config=None) -> List[Builder]:
        return list(self._builders)


@dataclasses.dataclass(frozen=True)
class TrainOpSpec:
    """Optimizer + hooks bundle (kept for API parity with reference
    adanet/subnetwork/generator.py:39-58; Builders may return one from
    build_optimizer)."""

    optimizer: Any
    chief_hooks: Tuple = ()
    hooks: Tuple = ()
