"""Storages: persistent stores of trained models + scores.

Reference: adanet/experimental/storages/storage.py +
in_memory_storage.py:26-58 (heap-ordered by score, lower is better).
"""

from __future__ import annotations

import abc
import heapq
import itertools
from typing import List


class ModelContainer(object):
    """(score, model, metrics) ordered by score (reference
    storages/storage.py ModelContainer)."""

    def __init__(self, score: float, model, metrics=None):
        self.score = score
        self.model = model
        self.metrics = dict(metrics or {})

    def __lt__(self, other):
        return self.score < other.score

    def __eq__(self, other):
        return isinstance(other, ModelContainer) and self.score == other.score


class Storage(abc.ABC):
    """Reference storages/storage.py Storage ABC."""

    @abc.abstractmethod
    def save_model(self, container: ModelContainer):
        ...

    @abc.abstractmethod
    def get_models(self) -> List:
        ...

    @abc.abstractmethod
    def get_best_models(self, num_models: int = 1) -> List:
        ...

    def get_model_metrics(self) -> List[dict]:
        return [c.metrics for c in self._containers()]

    @abc.abstractmethod
    def _containers(self) -> List[ModelContainer]:
        ...


class InMemoryStorage(Storage):
    """Heap-ordered in-memory storage (reference in_memory_storage.py:26-58)."""

    def __init__(self):
        self._heap: List = []
        self._tiebreak = itertools.count()

    def save_model(self, container: ModelContainer):
        heapq.heappush(self._heap, (container.score, next(self._tiebreak),
                                    container))

    def get_models(self) -> List:
        return [c.model for _, _, c in self._heap]

    def get_best_models(self, num_models: int = 1) -> List:
        return [c.model for _, _, c in heapq.nsmallest(
            num_models, self._heap)]

    def _containers(self) -> List[ModelContainer]:
        return [c for _, _, c in self._heap]
