"""Countdown timer for worker-wait timeouts (reference adanet/core/timer.py:25-45)."""

import time


class _CountDownTimer(object):

    def __init__(self, duration_secs: float):
        self._start = time.monotonic()
        self._duration_secs = duration_secs

    def secs_remaining(self) -> float:
        return max(0.0, self._duration_secs - (time.monotonic() - self._start))
