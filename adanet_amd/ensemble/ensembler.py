"""Ensembler / Ensemble abstract contracts.

Mirrors reference adanet/ensemble/ensembler.py:26-150 re-expressed for
define-by-run PyTorch: ``build_ensemble`` returns an ``EnsembleModule``
(an ``nn.Module``) rather than a namedtuple of graph tensors, and
``build_optimizer`` replaces ``build_train_op``.
"""

from __future__ import annotations

import abc
from typing import Any, List, Optional, Sequence

from torch import nn


class Ensemble(nn.Module, metaclass=abc.ABCMeta):
    """An ensemble of subnetworks (reference adanet/ensemble/ensembler.py:50-70).

    Concrete ensembles are ``nn.Module``s whose ``forward(features)`` returns
    the ensemble logits. They must expose:

    - ``subnetworks``: the list of member :class:`adanet_amd.subnetwork.Subnetwork`.
    - ``logits_from(sub_logits, sub_last_layers)``: combine precomputed member
      outputs (the engine uses this to reuse HBM-cached frozen logits — the
      MI355X analog of the reference's frozen-graph rebuild,
      adanet/core/iteration.py:569-572).
    """

    @property
    @abc.abstractmethod
    def subnetworks(self) -> List:
        """Member subnetworks, oldest first."""

    @abc.abstractmethod
    def logits_from(self, sub_logits, sub_last_layers):
        """Ensemble logits from the members' precomputed outputs."""


class TrainOpSpec(object):
    """Kept for API parity (reference adanet/ensemble/ensembler.py:26-47)."""

    def __init__(self, optimizer, chief_hooks=(), hooks=()):
        self.optimizer = optimizer
        self.chief_hooks = tuple(chief_hooks)
        self.hooks = tuple(hooks)


class Ensembler(abc.ABC):
    """Builds ensembles from groups of subnetworks.

    Reference: adanet/ensemble/ensembler.py:73-150.
    """

    @property
    @abc.abstractmethod
    def name(self) -> str:
        """This ensembler's name (used in scopes and checkpoints)."""

    @abc.abstractmethod
    def build_ensemble(self, subnetworks, previous_ensemble_subnetworks,
                       features, labels, logits_dimension, training,
                       previous_ensemble, device=None) -> Ensemble:
        """Builds and returns an Ensemble module combining ``subnetworks``."""

    def build_optimizer(self, ensemble: Ensemble, iteration: int = 0):
        """Optimizer over the ensemble's own (mixture) parameters, or None.

        Analog of reference ``Ensembler.build_train_op``
        (adanet/ensemble/ensembler.py:120-150). ``None`` means the ensemble
        has no trainable parameters of its own (e.g. MeanEnsembler) or the
        mixture weights should not be trained.
        """
        return None
