"""Ensemble API (reference: adanet/ensemble/__init__.py)."""

from adanet_amd.ensemble.ensembler import Ensemble
from adanet_amd.ensemble.ensembler import Ensembler
from adanet_amd.ensemble.ensembler import TrainOpSpec
from adanet_amd.ensemble.mean import MeanEnsemble
from adanet_amd.ensemble.mean import MeanEnsembler
from adanet_amd.ensemble.strategy import AllStrategy
from adanet_amd.ensemble.strategy import Candidate
from adanet_amd.ensemble.strategy import GrowStrategy
from adanet_amd.ensemble.strategy import SoloStrategy
from adanet_amd.ensemble.strategy import Strategy
from adanet_amd.ensemble.weighted import ComplexityRegularized
from adanet_amd.ensemble.weighted import ComplexityRegularizedEnsembler
from adanet_amd.ensemble.weighted import MixtureWeightType
from adanet_amd.ensemble.weighted import WeightedSubnetwork

__all__ = [
    "Ensemble",
    "Ensembler",
    "TrainOpSpec",
    "MeanEnsemble",
    "MeanEnsembler",
    "AllStrategy",
    "Candidate",
    "GrowStrategy",
    "SoloStrategy",
    "Strategy",
    "ComplexityRegularized",
    "ComplexityRegularizedEnsembler",
    "MixtureWeightType",
    "WeightedSubnetwork",
]
