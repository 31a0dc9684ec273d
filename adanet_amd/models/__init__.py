"""Canned model search spaces (reference: adanet/examples/ + research/)."""

from adanet_amd.models import simple_dnn

__all__ = ["simple_dnn"]
