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


class MultiGpuScheduler(Scheduler):
    """One worker per GPU, each pinned to its device via a per-GPU queue.

    Work units carrying a ``device`` attribute (TrainerPhase sets one when
    constructed with ``devices=[...]``) are routed to that GPU's queue;
    the rest are dealt round-robin. Each worker enters
    ``torch.cuda.device(i)`` so library calls inside the unit allocate and
    launch on its GPU, and units on the same GPU run serially (no
    intra-device contention) while different GPUs train concurrently —
    the ModelFlow analog of RoundRobinStrategy candidate parallelism.
    Falls back to serial execution when CUDA is unavailable.
    """

    def __init__(self, n_gpus: int = None):
        import torch
        if n_gpus is None:
            n_gpus = torch.cuda.device_count() if torch.cuda.is_available() \
                else 0
        self._n_gpus = int(n_gpus)

    def schedule(self, work_units: Iterator[WorkUnit]):
        self._run_batch(list(work_units))

    def schedule_phased(self, phased_work_units):
        for phase_units, parallel_ok in phased_work_units:
            if parallel_ok and self._n_gpus > 1:
                self._run_batch(list(phase_units))
            else:
                for wu in phase_units:
                    wu.execute()

    def _run_batch(self, units):
        import queue
        import threading

        import torch
        if not units:
            return
        if self._n_gpus <= 1:
            for wu in units:
                wu.execute()
            return
        queues = [queue.Queue() for _ in range(self._n_gpus)]
        rr = 0
        for wu in units:
            dev = getattr(wu, "device", None)
            idx = None
            if dev is not None:
                s = str(dev)
                if ":" in s:
                    try:
                        idx = int(s.split(":")[1]) % self._n_gpus
                    except ValueError:
                        idx = None
            if idx is None:
                idx = rr
                rr = (rr + 1) % self._n_gpus
            queues[idx].put(wu)
        errors = []

        def worker(gpu):
            try:
                with torch.cuda.device(gpu):
                    while True:
                        try:
                            wu = queues[gpu].get_nowait()
                        except queue.Empty:
                            return
                        wu.execute()
            except Exception as e:  # propagate after join
                errors.append(e)

        threads = [threading.Thread(target=worker, args=(g,))
                   for g in range(self._n_gpus)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        if errors:
            raise errors[0]
