"""Placement strategies: how candidates map onto the node's GPUs.

Reference: adanet/distributed/placement.py:31-320. The reference's axes
were parameter-server replication vs round-robin-over-candidates with
PS-group sharding. MI355X-native re-design (per SURVEY.md section 2.5
"MI355X mapping"):

  * ReplicationStrategy — every rank (GPU) builds and trains EVERY candidate
    data-parallel; gradients all-reduce per candidate per step over xGMI.
    The analog of reference placement.py:103-131.
  * RoundRobinStrategy — candidate subnetwork i is OWNED by rank
    (i mod world_size); each rank trains only its candidates (task
    parallelism — AdaNet's main scaling axis). The ensemble/mixture specs
    follow their new subnetwork's owner; the previous-best frozen candidate
    belongs to rank 0 (the analog of the reference's dedicated ensemble
    worker, placement.py:240-285). Ranks that own nothing this iteration
    idle until the iteration-end barrier (the reference's "dummy candidate",
    iteration.py:654-666, is unnecessary in define-by-run: there is no graph
    that must be non-empty).

Both strategies produce the same candidate list on every rank (Generators
must be deterministic) so iteration-end collectives line up by construction.
"""

from __future__ import annotations

import abc
from typing import List, Optional


class PlacementStrategy(abc.ABC):
    """Reference: adanet/distributed/placement.py:31-100."""

    def __init__(self):
        self.config = None  # set by the engine: RunConfig-like

    @property
    def world_size(self) -> int:
        return self.config.world_size if self.config is not None else 1

    @property
    def rank(self) -> int:
        return self.config.rank if self.config is not None else 0

    @abc.abstractmethod
    def should_build_subnetwork(self, num_subnetworks: int,
                                subnetwork_index: int) -> bool:
        """Whether this rank trains subnetwork `subnetwork_index`."""

    @abc.abstractmethod
    def should_build_ensemble(self, num_subnetworks: int) -> bool:
        """Whether this rank builds/trains candidate ensembles at all."""

    @abc.abstractmethod
    def should_train_subnetworks(self, num_subnetworks: int) -> bool:
        """Whether this rank participates in subnetwork training."""

    @abc.abstractmethod
    def subnetwork_owner(self, num_subnetworks: int,
                         subnetwork_index: int) -> int:
        """Rank that owns (authoritatively trains) the subnetwork."""

    @property
    @abc.abstractmethod
    def data_parallel(self) -> bool:
        """True when gradients must all-reduce across ranks each step."""


class ReplicationStrategy(PlacementStrategy):
    """Every rank trains every candidate, synchronous DP over xGMI.

    Reference: adanet/distributed/placement.py:103-131 (async-PS
    replication, re-done as synchronous RCCL data parallelism).
    """

    def should_build_subnetwork(self, num_subnetworks, subnetwork_index):
        return True

    def should_build_ensemble(self, num_subnetworks):
        return True

    def should_train_subnetworks(self, num_subnetworks):
        return True

    def subnetwork_owner(self, num_subnetworks, subnetwork_index):
        return 0  # all ranks hold identical weights; chief is authoritative

    @property
    def data_parallel(self):
        return True


class RoundRobinStrategy(PlacementStrategy):
    """Candidate i trains on rank (i mod world_size) — task parallelism.

    Reference: adanet/distributed/placement.py:134-320. drop_remainder
    mirrors the reference's semantics (placement.py:261-280): when True and
    there are more ranks than candidates, surplus ranks skip subnetwork
    training entirely instead of doubling up.
    """

    def __init__(self, drop_remainder: bool = False):
        super().__init__()
        self._drop_remainder = drop_remainder

    def subnetwork_owner(self, num_subnetworks, subnetwork_index):
        return subnetwork_index % self.world_size

    def should_build_subnetwork(self, num_subnetworks, subnetwork_index):
        return self.subnetwork_owner(num_subnetworks,
                                     subnetwork_index) == self.rank

    def should_build_ensemble(self, num_subnetworks):
        # Each rank builds the ensembles of the subnetworks it owns; rank 0
        # additionally owns the previous-best candidate.
        return True

    def should_train_subnetworks(self, num_subnetworks):
        if self._drop_remainder and self.rank >= num_subnetworks:
            return False
        return any(
            self.should_build_subnetwork(num_subnetworks, i)
            for i in range(num_subnetworks))

    @property
    def data_parallel(self):
        return False
