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
