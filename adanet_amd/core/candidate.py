"""_Candidate: an ensemble spec + EMA of its AdaNet loss.

Reference: adanet/core/candidate.py:28-138. The EMA (decay defaulting to
the Estimator's adanet_loss_decay=.9, reference estimator.py:615) is the
training-time proxy for candidate quality when no Evaluator is configured.
Host-side scalar math (K7 from SURVEY.md 2.9 — no kernel needed).
"""

from __future__ import annotations

import math
from typing import Optional


class _Candidate(object):

    def __init__(self, ensemble_spec, adanet_loss_decay: float = 0.9):
        if not 0.0 <= adanet_loss_decay < 1.0:
            raise ValueError("adanet_loss_decay must be in [0, 1)")
        self.ensemble_spec = ensemble_spec
        self._decay = adanet_loss_decay
        self._ema: Optional[float] = None

    @property
    def adanet_loss(self) -> float:
        """Current EMA of the candidate's adanet loss (NaN if diverged)."""
        if self._ema is None:
            return float("inf")
        return self._ema

    def update(self, adanet_loss: float):
        """assign_moving_average with zero-debias-free semantics
        (reference candidate.py:117-129). NaN PROPAGATES exactly like the
        reference's EMA arithmetic: once a candidate diverges its EMA stays
        NaN (selection maps it to -inf forever) — a later finite loss must
        NOT resurrect it."""
        v = float(adanet_loss)
        if math.isnan(v) or (self._ema is not None
                             and math.isnan(self._ema)):
            self._ema = float("nan")
            return
        if self._ema is None:
            self._ema = v
        else:
            self._ema = self._decay * self._ema + (1.0 - self._decay) * v
