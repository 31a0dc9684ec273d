"""Replay: deterministically re-run a previous AdaNet search.

Reference: adanet/replay/__init__.py:28-59. ``Config`` stores the sequence
of best-ensemble indices from a previous run; the engine consults it instead
of the Evaluator when selecting each iteration's winner (reference
adanet/core/estimator.py:1152-1157, 1433-1438).
"""

from typing import Optional, Sequence


class Config(object):
    """Replay configuration.

    Args:
        best_ensemble_indices: per-iteration index of the winning candidate
            ensemble from the run being replayed.
    """

    def __init__(self, best_ensemble_indices: Optional[Sequence[int]] = None):
        self._best_ensemble_indices = (
            list(best_ensemble_indices) if best_ensemble_indices is not None
            else None)

    @property
    def best_ensemble_indices(self):
        return self._best_ensemble_indices

    def get_best_ensemble_index(self, iteration_number: int) -> Optional[int]:
        """The stored winner for ``iteration_number``, or None past the end."""
        if (self._best_ensemble_indices is not None
                and iteration_number < len(self._best_ensemble_indices)):
            return self._best_ensemble_indices[iteration_number]
        return None


__all__ = ["Config"]
