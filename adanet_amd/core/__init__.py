"""Core engine (reference: adanet/core/__init__.py)."""

from adanet_amd.core.estimator import Estimator
from adanet_amd.core.evaluator import Evaluator, Objective
from adanet_amd.core.report_materializer import ReportMaterializer
from adanet_amd.core.summary import Summary

__all__ = ["Estimator", "Evaluator", "Objective", "ReportMaterializer",
           "Summary"]
