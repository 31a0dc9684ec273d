"""Ensemble models for ModelFlow.

Reference: adanet/experimental/keras/ensemble_model.py:26-86 —
EnsembleModel ABC, MeanEnsemble (average of submodel outputs),
WeightedEnsemble (Dense over stacked submodel outputs). The weighted
variant's Dense runs on HipLinear (fused MFMA GEMM).
"""

from __future__ import annotations

from typing import List, Sequence

import torch
from torch import nn

from adanet_amd.ops.linear import HipLinear


def _logits_of(model, x):
    m = model.module if hasattr(model, "module") else model
    out = m(x)
    return out[1] if isinstance(out, tuple) else out


class EnsembleModel(nn.Module):
    """Reference ensemble_model.py:26-50."""

    def __init__(self, submodels: Sequence, freeze_submodels: bool = True):
        super().__init__()
        self._submodels = list(submodels)
        mods = [m.module if hasattr(m, "module") else m for m in submodels]
        self._submodules = nn.ModuleList(mods)
        if freeze_submodels:
            for m in mods:
                for p in m.parameters():
                    p.requires_grad_(False)

    @property
    def submodels(self) -> List:
        return self._submodels


class MeanEnsemble(EnsembleModel):
    """Average of submodel outputs (reference ensemble_model.py:53-63)."""

    def forward(self, x):
        outs = [_logits_of(m, x) for m in self._submodels]
        return torch.stack(outs, dim=0).mean(dim=0)


class WeightedEnsemble(EnsembleModel):
    """Dense layer over concatenated submodel outputs
    (reference ensemble_model.py:66-86)."""

    def __init__(self, submodels: Sequence, output_units: int,
                 freeze_submodels: bool = True):
        super().__init__(submodels, freeze_submodels)
        self.output_units = output_units
        self.dense = HipLinear(output_units * len(self._submodels),
                               output_units)

    def forward(self, x):
        outs = [_logits_of(m, x) for m in self._submodels]
        stacked = torch.cat([o.reshape(o.shape[0], -1) for o in outs], dim=1)
        return self.dense(stacked)
