"""Evaluator: candidate selection over a shared eval dataset.

Reference: adanet/core/evaluator.py:31-140. After candidates finish
training, every candidate's metric (default: adanet_loss) is accumulated
over the same `steps` eval batches and the argmin (MINIMIZE) / argmax
(MAXIMIZE) wins. np.argmin/argmax semantics are kept — a NaN metric wins
argmin, surfacing divergence exactly like the reference.
"""

from __future__ import annotations

from typing import Callable, List, Optional, Sequence

import numpy as np


class Objective(object):
    """Minimize or maximize the eval metric (reference evaluator.py:34-60)."""

    MINIMIZE = "minimize"
    MAXIMIZE = "maximize"


class Evaluator(object):
    """Scores every candidate ensemble on `steps` shared batches of
    `input_fn` and selects argmin/argmax of `metric_name` (default
    adanet_loss; any head metric works — e.g. accuracy with MAXIMIZE).
    Reference adanet/core/evaluator.py:31-140."""

    def __init__(self, input_fn, steps: Optional[int] = None,
                 metric_name: str = "adanet_loss",
                 objective: str = Objective.MINIMIZE):
        if objective not in (Objective.MINIMIZE, Objective.MAXIMIZE):
            raise ValueError("objective must be 'minimize' or 'maximize'")
        self._input_fn = input_fn
        self._steps = steps
        self._metric_name = metric_name
        self._objective = objective

    @property
    def input_fn(self):
        return self._input_fn

    @property
    def steps(self):
        return self._steps

    @property
    def metric_name(self):
        return self._metric_name

    @property
    def objective(self):
        return self._objective

    def evaluate(self, metric_fns: Sequence[Callable],
                 input_iter=None) -> List[float]:
        """Accumulates each candidate's mean metric over the eval batches.

        Args:
            metric_fns: per-candidate callables mapping (features, labels) ->
                float metric for that batch (the engine wires these to the
                candidate's adanet_loss computed with HBM-cached frozen
                logits).
            input_iter: optional pre-built iterator (engine supplies one so
                all candidates see identical batches, mirroring the
                reference's single-Session shared-batch loop,
                evaluator.py:124-137).

        Returns:
            Per-candidate mean metric values.
        """
        it = input_iter if input_iter is not None else iter(self._input_fn())
        sums = [0.0] * len(metric_fns)
        count = 0
        step = 0
        while self._steps is None or step < self._steps:
            try:
                features, labels = next(it)
            except StopIteration:
                break
            for i, fn in enumerate(metric_fns):
                # Metric fns may return device tensors; keep accumulation
                # on-device and sync once per candidate at the end.
                sums[i] = sums[i] + fn(features, labels)
            count += 1
            step += 1
        if count == 0:
            return [float("nan")] * len(metric_fns)
        return [float(s) / count for s in sums]

    def best_index(self, values: Sequence[float]) -> int:
        arr = np.asarray(values, dtype=np.float64)
        if self._objective == Objective.MINIMIZE:
            return int(np.argmin(arr))
        return int(np.argmax(arr))
