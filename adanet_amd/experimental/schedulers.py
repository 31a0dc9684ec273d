"""Schedulers: execute the controller's work units.

Reference: adanet/experimental/schedulers/{scheduler.py,
in_process_scheduler.py:27-37}. Plus an MI355X-native MultiGpuScheduler:
independent work units (candidate trainings) dispatch round-robin onto the
node's GPUs, one subprocess per GPU — the ModelFlow analog of
RoundRobinStrategy candidate parallelism. Work units within one GPU run
serially.
"""

from __future__ import annotations

import abc
from typing import Iterator

from adanet_amd.experimental.work_units import WorkUnit


class Scheduler(abc.ABC):

    @abc.abstractmethod
    def schedule(self, work_units: Iterator[WorkUnit]):
        ...


class InProcessScheduler(Scheduler):
    """Serial in-process execution (reference in_process_scheduler.py:27-37)."""

    def schedule(self, work_units: Iterator[WorkUnit]):
        for wu in work_units:
            wu.execute()


class ThreadedScheduler(Scheduler):
    """Executes up to ``max_workers`` work units concurrently.

    The MI355X-native scheduler for multi-GPU ModelFlow: python threads
    release the GIL while device work runs, so work units whose models live
    on different GPUs (TrainerPhase(devices=[...]) round-robins them)
    train concurrently — the ModelFlow analog of RoundRobinStrategy
    candidate parallelism. Phase boundaries are barriers: a phase's work
    units all finish before the next phase's begin (phases consume their
    predecessor's storage).
    """

    def __init__(self, max_workers: int = 8):
        self._max_workers = max(1, int(max_workers))

    def schedule(self, work_units: Iterator[WorkUnit]):
        # Without phase structure every unit is assumed independent.
        self._run_batch(list(work_units))

    def schedule_phased(self, phased_work_units):
        """phased_work_units: iterable of (work_units, parallel_ok) pairs
        (ModelSearch passes SequentialController.phased_work_units())."""
        for phase_units, parallel_ok in phased_work_units:
            if parallel_ok:
                self._run_batch(list(phase_units))
            else:
                for wu in phase_units:
                    wu.execute()

    def _run_batch(self, units):
        import concurrent.futures
        if not units:
            return
        with concurrent.futures.ThreadPoolExecutor(
                max_workers=self._max_workers) as pool:
            futures = [pool.submit(wu.execute) for wu in units]
            for f in futures:
                f.result()  # propagate exceptions
