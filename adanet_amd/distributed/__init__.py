"""Distributed layer (reference: adanet/distributed/__init__.py)."""

from adanet_amd.distributed import comm
from adanet_amd.distributed.placement import PlacementStrategy
from adanet_amd.distributed.placement import ReplicationStrategy
from adanet_amd.distributed.placement import RoundRobinStrategy

__all__ = [
    "comm",
    "PlacementStrategy",
    "ReplicationStrategy",
    "RoundRobinStrategy",
]
