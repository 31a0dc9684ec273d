"""Per-subnetwork / per-ensemble / per-iteration eval metric stores.

Reference: adanet/core/eval_metrics.py:41-427. The reference builds TF
metric-op graphs whose "best" variants dynamically mux
tf.stack(values)[best_candidate_index]; define-by-run needs no graph: the
stores accumulate streaming metric values per candidate and the iteration
store selects the best candidate's values at read time (K11 — host-side
mux). Architecture is exported as a text metric
(reference _architecture_as_metric, eval_metrics.py:227-264); replay
indices as best_ensemble_index_<i> (eval_metrics.py:332-350).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence


class _MeanAccumulator(object):

    def __init__(self):
        self.total = 0.0
        self.count = 0

    def update(self, value: float, n: int = 1):
        self.total += float(value) * n
        self.count += n

    @property
    def value(self) -> float:
        return self.total / self.count if self.count else float("nan")


class _EvalMetricsStore(object):
    """Accumulates streaming means of named metrics
    (reference eval_metrics.py:41-68)."""

    def __init__(self):
        self._metrics: Dict[str, _MeanAccumulator] = {}

    def update(self, metrics: Dict[str, float], n: int = 1):
        for k, v in metrics.items():
            self._metrics.setdefault(k, _MeanAccumulator()).update(v, n)

    def result(self) -> Dict[str, float]:
        return {k: acc.value for k, acc in self._metrics.items()}


class _SubnetworkMetrics(_EvalMetricsStore):
    """Reference eval_metrics.py:71-213."""


class _EnsembleMetrics(_EvalMetricsStore):
    """Reference eval_metrics.py:215-264 (adds the architecture text)."""

    def __init__(self, architecture_json: str = ""):
        super().__init__()
        self.architecture = architecture_json

    def result(self):
        out = super().result()
        if self.architecture:
            out["architecture/adanet/ensembles"] = self.architecture
        return out


class _IterationMetrics(object):
    """Muxes the best candidate's metrics (reference eval_metrics.py:267-427)."""

    def __init__(self, iteration_number: int,
                 ensemble_metrics: Sequence[_EnsembleMetrics],
                 subnetwork_metrics: Sequence[_SubnetworkMetrics] = (),
                 replay_indices: Optional[List[int]] = None):
        self.iteration_number = iteration_number
        self.ensemble_metrics = list(ensemble_metrics)
        self.subnetwork_metrics = list(subnetwork_metrics)
        self.replay_indices = list(replay_indices or [])

    def best_eval_metrics(self, best_index: int) -> Dict[str, float]:
        """The winning candidate's metrics + iteration + replay indices
        (reference best_eval_metrics_tuple, eval_metrics.py:306-408)."""
        out = {}
        if 0 <= best_index < len(self.ensemble_metrics):
            out.update(self.ensemble_metrics[best_index].result())
        out["iteration"] = self.iteration_number
        for i, idx in enumerate(self.replay_indices):
            out["best_ensemble_index_%d" % i] = idx
        return out
