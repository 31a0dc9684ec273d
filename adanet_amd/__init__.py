"""adanet_amd — an MI355X-native AdaNet adaptive-ensemble AutoML engine.

Public API mirrors the reference package facade (adanet/__init__.py:21-59):
Estimator, AutoEnsembleEstimator, AutoEnsembleSubestimator, Evaluator,
ReportMaterializer, Summary, Ensemble, MixtureWeightType, WeightedSubnetwork,
Subnetwork, plus the subpackages subnetwork / ensemble / distributed /
replay. Compute path: PyTorch-ROCm + hand-written HIP/CDNA4 kernels
(adanet_amd/csrc, gfx950) + RCCL over xGMI.
"""

from adanet_amd import distributed
from adanet_amd import ensemble
from adanet_amd import head
from adanet_amd import hooks
from adanet_amd import models
from adanet_amd import ops
from adanet_amd import replay
from adanet_amd import serving
from adanet_amd import subnetwork
from adanet_amd.autoensemble.common import AutoEnsembleSubestimator
from adanet_amd.autoensemble.estimator import AutoEnsembleEstimator
from adanet_amd.config import RunConfig
from adanet_amd.core.estimator import Estimator
from adanet_amd.core.estimator import NanLossDuringTrainingError
from adanet_amd.core.evaluator import Evaluator
from adanet_amd.core.report_materializer import ReportMaterializer
from adanet_amd.core.summary import Summary
from adanet_amd.ensemble.ensembler import Ensemble
from adanet_amd.ensemble.weighted import MixtureWeightType
from adanet_amd.ensemble.weighted import WeightedSubnetwork
from adanet_amd.subnetwork.generator import Subnetwork
from adanet_amd.version import __version__

__all__ = [
    "AutoEnsembleEstimator",
    "AutoEnsembleSubestimator",
    "Ensemble",
    "Estimator",
    "Evaluator",
    "MixtureWeightType",
    "NanLossDuringTrainingError",
    "ReportMaterializer",
    "RunConfig",
    "Subnetwork",
    "Summary",
    "WeightedSubnetwork",
    "distributed",
    "ensemble",
    "head",
    "hooks",
    "models",
    "ops",
    "replay",
    "serving",
    "subnetwork",
    "__version__",
]
