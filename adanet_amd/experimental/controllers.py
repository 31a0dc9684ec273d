"""Controllers: orchestrate phases into a work-unit stream.

Reference: adanet/experimental/controllers/{controller.py,
sequential_controller.py:26-49}.
"""

from __future__ import annotations

import abc
from typing import Iterator, List, Sequence

from adanet_amd.experimental.phases import Phase
from adanet_amd.experimental.work_units import WorkUnit


class Controller(abc.ABC):

    @abc.abstractmethod
    def work_units(self) -> Iterator[WorkUnit]:
        ...

    @abc.abstractmethod
    def get_best_models(self, num_models: int = 1) -> List:
        ...


class SequentialController(Controller):
    """Chains phases in order (reference sequential_controller.py:26-49)."""

    def __init__(self, phases: Sequence[Phase]):
        self._phases = list(phases)
        prev = None
        for phase in self._phases:
            if prev is not None:
                phase.set_previous(prev)
            prev = phase

    def work_units(self):
        for phase in self._phases:
            for wu in phase.work_units():
                yield wu

    def phased_work_units(self):
        """Yields (work_units, parallel_ok) per phase: a phase's units are
        mutually independent unless the phase says otherwise (RepeatPhase
        chains repetitions); phases are barriers."""
        for phase in self._phases:
            yield phase.work_units(), getattr(phase, "parallel_ok", True)

    def get_best_models(self, num_models: int = 1):
        return self._phases[-1].get_best_models(num_models)
