"""Ensemble candidate strategies.

Mirrors reference adanet/ensemble/strategy.py:26-117 semantics exactly:
a Strategy maps the iteration's subnetwork Builders to a list of Candidate
ensembles (subsets of builders + previous-ensemble builders).
"""

from __future__ import annotations

import abc
import dataclasses
from typing import List, Optional, Sequence


@dataclasses.dataclass(frozen=True)
class Candidate:
    """An ensemble candidate: which builders to combine.

    Reference: adanet/ensemble/strategy.py:26-48.
    """

    name: str
    subnetwork_builders: tuple
    previous_ensemble_subnetwork_builders: tuple

    def __init__(self, name, subnetwork_builders,
                 previous_ensemble_subnetwork_builders):
        object.__setattr__(self, "name", name)
        object.__setattr__(self, "subnetwork_builders",
                           tuple(subnetwork_builders))
        object.__setattr__(
            self, "previous_ensemble_subnetwork_builders",
            tuple(previous_ensemble_subnetwork_builders or ()))


class Strategy(abc.ABC):
    """Generates ensemble candidates from the iteration's builders.

    Reference: adanet/ensemble/strategy.py:51-76.
    """

    @abc.abstractmethod
    def generate_ensemble_candidates(
        self, subnetwork_builders: Sequence,
        previous_ensemble_subnetwork_builders: Optional[Sequence],
    ) -> List[Candidate]:
        """Returns ensemble Candidates to explore this iteration."""


class SoloStrategy(Strategy):
    """Each subnetwork as its own standalone ensemble ("pure" subnetwork
    performance; useful for comparing ensembling value).

    Reference: adanet/ensemble/strategy.py:79-94.
    """

    def generate_ensemble_candidates(self, subnetwork_builders,
                                     previous_ensemble_subnetwork_builders):
        return [
            Candidate("{}_solo".format(b.name), [b], None)
            for b in subnetwork_builders
        ]


class GrowStrategy(Strategy):
    """Greedily grows: each candidate = one new subnetwork + all previous
    ensemble subnetworks. The AdaNet default.

    Reference: adanet/ensemble/strategy.py:97-106; set as the default
    strategy in adanet/core/estimator.py:754-756.
    """

    def generate_ensemble_candidates(self, subnetwork_builders,
                                     previous_ensemble_subnetwork_builders):
        return [
            Candidate("{}_grow".format(b.name), [b],
                      previous_ensemble_subnetwork_builders)
            for b in subnetwork_builders
        ]


class AllStrategy(Strategy):
    """One candidate ensembling every new subnetwork + all previous.

    Reference: adanet/ensemble/strategy.py:109-117.
    """

    def generate_ensemble_candidates(self, subnetwork_builders,
                                     previous_ensemble_subnetwork_builders):
        return [
            Candidate("all", subnetwork_builders,
                      previous_ensemble_subnetwork_builders)
        ]
