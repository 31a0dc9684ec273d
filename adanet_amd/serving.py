"""Serving: load an exported AdaNet ensemble for inference.

The analog of reference SavedModel export/serving
(adanet/core/estimator.py:1090-1146 + iteration.py:1111-1186 export
muxing): export_saved_model() writes a self-contained saved_model.pt
bundle; load_ensemble() reconstructs the frozen best ensemble given the
same (deterministic) subnetwork generator.
"""

from __future__ import annotations

import os
from typing import Optional

import torch


class ServableEnsemble(torch.nn.Module):
    """Frozen best ensemble + prediction helpers."""

    def __init__(self, estimator, head):
        super().__init__()
        self._est = estimator
        self._head = head
        self._ens = None

    def forward(self, features):
        ens, _ = self._est._load_frozen_best()
        return ens(features)

    def predict(self, features):
        with torch.no_grad():
            return self._head.predictions(self(features))


def load_ensemble(export_dir: str, subnetwork_generator, head,
                  device: Optional[str] = None) -> ServableEnsemble:
    """Loads a saved_model.pt bundle produced by export_saved_model().

    Args:
        export_dir: the timestamped export directory.
        subnetwork_generator: the SAME (deterministic) generator used for
            training — builders are re-invoked to reconstruct module
            structure, then weights load from the bundle (the reference
            replays architectures the same way, estimator.py:1785-1882).
        head: the head used in training.
        device: target device (default: cuda if available).
    """
    from adanet_amd.config import RunConfig
    from adanet_amd.core.estimator import Estimator

    path = os.path.join(export_dir, "saved_model.pt")
    payload = torch.load(path, map_location="cpu", weights_only=False)
    if payload.get("format") != "adanet_amd.v1":
        raise ValueError("Unknown export format in %s" % path)
    est = Estimator.__new__(Estimator)
    # Minimal state needed by _load_frozen_best/_rebuild_previous_ensemble.
    est._head = head
    est._subnetwork_generator = subnetwork_generator
    est._report_materializer = None
    est._ensemblers = []
    from adanet_amd.ensemble import ComplexityRegularizedEnsembler
    est._ensemblers = [ComplexityRegularizedEnsembler()]
    est._config = RunConfig(device=device)
    est._device = est._config.resolve_device()
    est._architectures = {
        int(k): v for k, v in payload["architectures"].items()
    }
    est._frozen_states = payload["frozen_states"]
    est._best_ensemble_state = payload["ensemble_state"]
    est._replay_indices = list(payload.get("replay_indices", []))
    est._iteration_number = max(est._architectures) + 1 if (
        est._architectures) else 0
    est._global_step = 0
    est._frozen_best_cache = None
    return ServableEnsemble(est, head)
