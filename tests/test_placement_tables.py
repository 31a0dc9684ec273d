"""Exhaustive placement truth tables (reference placement_test.py:66-548
pattern: full grids over world x candidates x drop_remainder)."""

import pytest

from adanet_amd.distributed.placement import (ReplicationStrategy,
                                              RoundRobinStrategy)


def _patch(strat, world, rank, monkeypatch):
    monkeypatch.setattr(type(strat), "world_size",
                        property(lambda self: world))
    monkeypatch.setattr(type(strat), "rank", property(lambda self: rank))
    return strat


@pytest.mark.parametrize("world", [1, 2, 3, 4, 8])
@pytest.mark.parametrize("num_subnetworks", [1, 2, 3, 5, 8, 16])
def test_replication_builds_everything(world, num_subnetworks, monkeypatch):
    for rank in range(world):
        s = _patch(ReplicationStrategy(), world, rank, monkeypatch)
        for i in range(num_subnetworks):
            assert s.should_build_subnetwork(num_subnetworks, i)
        assert s.should_build_ensemble(num_subnetworks)
        assert s.should_train_subnetworks(num_subnetworks)
        assert s.data_parallel


@pytest.mark.parametrize("world", [1, 2, 3, 4, 8])
@pytest.mark.parametrize("num_subnetworks", [1, 2, 3, 5, 8, 16])
@pytest.mark.parametrize("drop_remainder", [False, True])
def test_round_robin_partition(world, num_subnetworks, drop_remainder,
                               monkeypatch):
    """Every candidate owned by exactly ONE rank; owners = i %% world;
    drop_remainder idles surplus ranks."""
    owners = {}
    for rank in range(world):
        s = _patch(RoundRobinStrategy(drop_remainder=drop_remainder), world,
                   rank, monkeypatch)
        assert not s.data_parallel
        for i in range(num_subnetworks):
            assert s.subnetwork_owner(num_subnetworks, i) == i % world
            if s.should_build_subnetwork(num_subnetworks, i):
                assert i not in owners, "double ownership"
                owners[i] = rank
        trains = s.should_train_subnetworks(num_subnetworks)
        if drop_remainder and rank >= num_subnetworks:
            assert not trains
        else:
            # trains iff this rank owns at least one candidate
            assert trains == any(i % world == rank
                                 for i in range(num_subnetworks))
    assert sorted(owners) == list(range(num_subnetworks))
    # weak-scaling invariant of the bench: with 2*world candidates every
    # rank owns exactly 2
    if num_subnetworks == 2 * world:
        from collections import Counter
        per_rank = Counter(owners.values())
        assert all(per_rank[r] == 2 for r in range(world))
