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


class AUCAccumulator(object):
    """Streaming thresholded AUC / precision / recall for binary heads.

    Mirrors tf.metrics.auc's bucketed accumulation (the reference wires
    tf.estimator binary heads whose eval dict includes auc/auc_pr,
    reference adanet/core/eval_metrics.py metric-op passthrough): scores
    in [0,1] are bucketed into `num_thresholds` bins per class; suffix
    sums give TP/FP at every threshold and the ROC is integrated by
    trapezoid. On GPU the histogram is the native binary_histogram kernel
    (csrc/reduce.hip); on CPU a torch bincount.
    """

    def __init__(self, num_thresholds: int = 200):
        self.T = int(num_thresholds)
        self._hist = None  # lazily placed [2, T] int on first update

    def update(self, scores, labels):
        import torch
        from adanet_amd.ops import _extension
        scores = scores.detach().reshape(-1).float()
        labels = labels.detach().reshape(-1).long()
        if self._hist is None:
            self._hist = torch.zeros((2, self.T), dtype=torch.int32,
                                     device=scores.device)
        if scores.is_cuda:
            ext = _extension.require()
            ext.binary_histogram(scores.contiguous(), labels.contiguous(),
                                 self._hist)
        else:
            b = (scores.clamp(0, 1) * self.T).long().clamp(max=self.T - 1)
            pos = torch.bincount(b[labels != 0], minlength=self.T)
            neg = torch.bincount(b[labels == 0], minlength=self.T)
            self._hist[0] += pos.to(torch.int32)
            self._hist[1] += neg.to(torch.int32)

    def _counts(self):
        import torch
        h = self._hist.cpu().long()
        pos, neg = h[0], h[1]
        # TP/FP predicting positive at threshold = bucket lower edge
        # (suffix-inclusive), thresholds 0..T-1 plus the all-negative end.
        tp = torch.flip(torch.cumsum(torch.flip(pos, [0]), 0), [0])
        fp = torch.flip(torch.cumsum(torch.flip(neg, [0]), 0), [0])
        return tp, fp, int(pos.sum()), int(neg.sum())

    def value(self) -> Dict[str, float]:
        if self._hist is None:
            return {"auc": float("nan"), "precision": float("nan"),
                    "recall": float("nan")}
        tp, fp, p, n = self._counts()
        if p == 0 or n == 0:
            auc = float("nan")
        else:
            tpr = tp.double() / p
            fpr = fp.double() / n
            # thresholds descend left->right along the ROC; append (0,0)
            import torch
            tpr = torch.cat([tpr, torch.zeros(1, dtype=torch.float64)])
            fpr = torch.cat([fpr, torch.zeros(1, dtype=torch.float64)])
            auc = float(torch.trapz(tpr.flip(0), fpr.flip(0)))
        mid = self.T // 2
        tp5, fp5 = int(tp[mid]), int(fp[mid])
        fn5 = p - tp5
        precision = tp5 / (tp5 + fp5) if (tp5 + fp5) else float("nan")
        recall = tp5 / (tp5 + fn5) if (tp5 + fn5) else float("nan")
        return {"auc": auc, "precision": precision, "recall": recall}


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
