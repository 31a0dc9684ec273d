"""Auto-ensembling (reference: adanet/autoensemble/__init__.py)."""

from adanet_amd.autoensemble.common import AutoEnsembleSubestimator
from adanet_amd.autoensemble.estimator import AutoEnsembleEstimator

__all__ = ["AutoEnsembleEstimator", "AutoEnsembleSubestimator"]
