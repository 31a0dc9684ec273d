"""Search-space API (reference: adanet/subnetwork/__init__.py)."""

from adanet_amd.subnetwork.generator import Builder
from adanet_amd.subnetwork.generator import Generator
from adanet_amd.subnetwork.generator import SimpleGenerator
from adanet_amd.subnetwork.generator import Subnetwork
from adanet_amd.subnetwork.generator import SubnetworkModule
from adanet_amd.subnetwork.generator import TrainOpSpec
from adanet_amd.subnetwork.report import MaterializedReport
from adanet_amd.subnetwork.report import Report

__all__ = [
    "Builder",
    "Generator",
    "SimpleGenerator",
    "Subnetwork",
    "SubnetworkModule",
    "TrainOpSpec",
    "MaterializedReport",
    "Report",
]
