"""MeanEnsembler: uniform average of member logits.

Reference: adanet/ensemble/mean.py:27-135.
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import torch
from torch import nn

from adanet_amd.ensemble.ensembler import Ensemble, Ensembler
from adanet_amd.subnetwork.generator import Subnetwork


class MeanEnsemble(Ensemble):
    """Mean-of-logits ensemble (reference adanet/ensemble/mean.py:27-53).

    When ``add_mean_last_layer_predictions`` the forward also records the
    mean last layer under ``self.last_mean_last_layer`` (the reference adds a
    ``mean_last_layer`` key to predictions, mean.py:40-52).
    """

    def __init__(self, subnetworks: Sequence[Subnetwork],
                 add_mean_last_layer_predictions: bool = False):
        super().__init__()
        self._subnetworks = list(subnetworks)
        # Register member modules so .to()/.state_dict() see them.
        self._modules_list = nn.ModuleList([s.module for s in subnetworks])
        self.add_mean_last_layer_predictions = add_mean_last_layer_predictions
        self.last_mean_last_layer = None

    @property
    def subnetworks(self) -> List[Subnetwork]:
        return self._subnetworks

    def forward(self, features):
        sub_logits, sub_last = [], []
        for s in self._subnetworks:
            last_layer, logits = s(features)
            sub_logits.append(logits)
            sub_last.append(last_layer)
        return self.logits_from(sub_logits, sub_last)

    def logits_from(self, sub_logits, sub_last_layers):
        logits = torch.stack(sub_logits, dim=0).mean(dim=0)
        if self.add_mean_last_layer_predictions:
            try:
                self.last_mean_last_layer = torch.stack(
                    sub_last_layers, dim=0).mean(dim=0)
            except RuntimeError as e:
                # Mirrors the reference's error when last-layer shapes differ
                # (mean.py:117-126).
                raise ValueError(
                    "Shapes of last_layers must be identical to use "
                    "add_mean_last_layer_predictions") from e
        return logits

    def complexity_regularization(self) -> torch.Tensor:
        dev = "cpu"
        for s in self._subnetworks:
            for p in s.module.parameters():
                dev = p.device
                break
            break
        return torch.zeros((), device=dev)


class MeanEnsembler(Ensembler):
    """Ensembler producing MeanEnsembles (reference adanet/ensemble/mean.py:56-135)."""

    def __init__(self, name: Optional[str] = None,
                 add_mean_last_layer_predictions: bool = False):
        self._name = name
        self._add_mean_last_layer_predictions = add_mean_last_layer_predictions

    @property
    def name(self) -> str:
        return self._name or "mean"

    def build_ensemble(self, subnetworks, previous_ensemble_subnetworks,
                       features, labels, logits_dimension, training,
                       previous_ensemble, device=None) -> MeanEnsemble:
        members = []
        prev = list(previous_ensemble_subnetworks or [])
        if prev and previous_ensemble is not None:
            for s in previous_ensemble.subnetworks:
                if s in prev:
                    members.append(s)
        members.extend(subnetworks)
        ens = MeanEnsemble(
            members,
            add_mean_last_layer_predictions=(
                self._add_mean_last_layer_predictions))
        if device is not None:
            ens = ens.to(device)
        return ens

    def build_optimizer(self, ensemble, iteration: int = 0):
        # Mean ensembles have no mixture parameters (reference mean.py:128-135
        # returns tf.no_op()).
        return None
