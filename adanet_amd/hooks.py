"""Training hooks: the define-by-run analog of tf SessionRunHooks.

The reference threads SessionRunHooks everywhere (TrainOpSpec hook fields,
adanet/subnetwork/generator.py:39-58; estimator train hooks). Here a hook
is a small object with begin/before_step/after_step/end callbacks invoked
by Estimator.train around the lockstep iteration steps. Builders may attach
hooks by returning a TrainOpSpec from build_optimizer; chief_hooks run only
on rank 0.
"""

from __future__ import annotations

from typing import Optional


class TrainHook(object):
    """Base hook; override any subset."""

    def begin(self, estimator=None, iteration=None):
        pass

    def before_step(self, global_step: int):
        pass

    def after_step(self, global_step: int):
        pass

    def end(self, estimator=None):
        pass


class EveryNSteps(TrainHook):
    """Calls fn(global_step) every n steps (a convenience like
    tf.train.StepCounterHook-style cadence hooks)."""

    def __init__(self, n: int, fn):
        self._n = max(1, int(n))
        self._fn = fn

    def after_step(self, global_step: int):
        if global_step % self._n == 0:
            self._fn(global_step)


class StopAfterSteps(TrainHook):
    """Requests a stop after n steps of the current train() call (the
    reference's _StopAfterTrainingHook flavor, estimator.py:50-87)."""

    def __init__(self, n: int):
        self._n = int(n)
        self._seen = 0
        self.should_stop = False

    def after_step(self, global_step: int):
        self._seen += 1
        if self._seen >= self._n:
            self.should_stop = True
