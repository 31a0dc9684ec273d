"""ComplexityRegularizedEnsembler — the AdaNet objective (Eq. 4).

MI355X-native re-implementation of reference adanet/ensemble/weighted.py:
the ensemble logits are

    F(x) = b + sum_j mix(w_j, h_j(x))

where ``mix`` is elementwise scale (SCALAR), per-class scale (VECTOR) or
``last_layer @ W_j`` (MATRIX) — reference weighted.py:400-454, 545-561 —
and the complexity regularization term is

    sum_j (lambda * r(h_j) + beta) * ||w_j||_1        (weighted.py:563-604)

On GPU, the scalar/vector sum-of-weighted-logits runs as one fused HIP
kernel over the J cached logit buffers (see adanet_amd/csrc/mixer.hip and
adanet_amd/ops/mixer.py) instead of J separate multiply-adds; frozen member
logits come from the iteration's HBM logit cache so frozen subnetworks are
never re-run for the mixer (north star: frozen logits cached in 288 GB HBM).
"""

from __future__ import annotations

import math
from typing import List, Optional, Sequence

import torch
from torch import nn

from adanet_amd.ensemble.ensembler import Ensemble, Ensembler
from adanet_amd.subnetwork.generator import Subnetwork


class MixtureWeightType(object):
    """Mixture weight types (reference adanet/ensemble/weighted.py:135-148)."""

    SCALAR = "scalar"
    VECTOR = "vector"
    MATRIX = "matrix"


class WeightedSubnetwork(nn.Module):
    """A subnetwork paired with its learned mixture weight.

    Reference: adanet/ensemble/weighted.py:43-85 (namedtuple) — here an
    ``nn.Module`` owning the mixture-weight parameter. The wrapped
    subnetwork's parameters may be frozen (``requires_grad=False``) when it
    comes from a previous iteration; the mixture weight always requires grad
    (trained by the ensembler's optimizer).
    """

    def __init__(self, subnetwork: Subnetwork, weight: torch.Tensor,
                 mixture_weight_type: str, iteration_number: int = 0,
                 builder_name: str = ""):
        super().__init__()
        self.subnetwork = subnetwork
        self.mixture_weight_type = mixture_weight_type
        self.iteration_number = iteration_number
        self.builder_name = builder_name or subnetwork.name
        self.weight = nn.Parameter(weight)

    @property
    def name(self) -> str:
        return self.builder_name

    def mix(self, last_layer: torch.Tensor,
            logits: torch.Tensor) -> torch.Tensor:
        """Weighted logits contribution (reference weighted.py:427-454)."""
        t = self.mixture_weight_type
        if t == MixtureWeightType.MATRIX:
            # last_layer @ W, supporting rank-3 last layers by flattening the
            # middle dims (reference weighted.py:434-451).
            ll = last_layer
            if ll.dim() > 2:
                ll = ll.reshape(ll.shape[0], -1)
            return ll.to(self.weight.dtype) @ self.weight
        # SCALAR / VECTOR: elementwise on logits.
        return logits * self.weight


class ComplexityRegularized(Ensemble):
    """The ensemble produced by :class:`ComplexityRegularizedEnsembler`.

    Reference: adanet/ensemble/weighted.py:88-132.
    """

    def __init__(self, weighted_subnetworks: Sequence[WeightedSubnetwork],
                 bias: Optional[torch.Tensor], adanet_lambda: float,
                 adanet_beta: float, use_bias: bool):
        super().__init__()
        self.weighted_subnetworks = nn.ModuleList(weighted_subnetworks)
        self.adanet_lambda = float(adanet_lambda)
        self.adanet_beta = float(adanet_beta)
        self.use_bias = use_bias
        if use_bias and bias is not None:
            self.bias = nn.Parameter(bias)
        else:
            self.register_parameter("bias", None)

    @property
    def subnetworks(self) -> List[Subnetwork]:
        return [ws.subnetwork for ws in self.weighted_subnetworks]

    def mixture_parameters(self):
        params = [ws.weight for ws in self.weighted_subnetworks]
        if self.bias is not None:
            params.append(self.bias)
        return params

    def forward(self, features):
        sub_logits, sub_last = [], []
        for ws in self.weighted_subnetworks:
            last_layer, logits = ws.subnetwork(features)
            sub_logits.append(logits)
            sub_last.append(last_layer)
        return self.logits_from(sub_logits, sub_last)

    def logits_from(self, sub_logits, sub_last_layers):
        """Combine precomputed member outputs into ensemble logits.

        Scalar/vector mixtures dispatch to the fused weighted-sum HIP kernel
        (K5) when every member logit is a CUDA tensor of identical shape;
        matrix mixtures fall back to per-member GEMMs.
        """
        types = {ws.mixture_weight_type for ws in self.weighted_subnetworks}
        if (types <= {MixtureWeightType.SCALAR, MixtureWeightType.VECTOR}
                and len(sub_logits) > 0):
            from adanet_amd.ops import mixer
            weights = [ws.weight for ws in self.weighted_subnetworks]
            return mixer.weighted_sum_logits(sub_logits, weights, self.bias)
        # Mixed / matrix path.
        total = None
        for ws, ll, lg in zip(self.weighted_subnetworks, sub_last_layers,
                              sub_logits):
            contrib = ws.mix(ll, lg)
            total = contrib if total is None else total + contrib
        if self.bias is not None:
            total = total + self.bias
        return total

    def complexity_regularization(self) -> torch.Tensor:
        """sum_j (lambda*r_j + beta) * ||w_j||_1 (reference weighted.py:563-604)."""
        total = None
        for ws in self.weighted_subnetworks:
            coef = (self.adanet_lambda * float(ws.subnetwork.complexity)
                    + self.adanet_beta)
            if coef == 0.0:
                continue
            term = coef * ws.weight.abs().sum()
            total = term if total is None else total + term
        if total is None:
            dev = (self.weighted_subnetworks[0].weight.device
                   if len(self.weighted_subnetworks) else "cpu")
            total = torch.zeros((), device=dev)
        return total


class ComplexityRegularizedEnsembler(Ensembler):
    """AdaNet's ensembler (reference adanet/ensemble/weighted.py:150-617).

    Constructor signature mirrors the reference exactly
    (weighted.py:223-248). ``optimizer`` is a callable
    ``params -> torch.optim.Optimizer`` (or None for untrained mixture
    weights, the analog of the reference's ``tf.no_op()`` default).
    """

    def __init__(self, optimizer=None,
                 mixture_weight_type: str = MixtureWeightType.SCALAR,
                 mixture_weight_initializer=None,
                 warm_start_mixture_weights: bool = False,
                 model_dir: Optional[str] = None,
                 adanet_lambda: float = 0.,
                 adanet_beta: float = 0.,
                 use_bias: bool = False,
                 name: Optional[str] = None):
        if warm_start_mixture_weights and model_dir is None:
            raise ValueError("model_dir cannot be None when "
                             "warm_start_mixture_weights is True.")
        self._optimizer = optimizer
        self._mixture_weight_type = mixture_weight_type
        self._mixture_weight_initializer = mixture_weight_initializer
        self._warm_start_mixture_weights = warm_start_mixture_weights
        self._model_dir = model_dir
        self._adanet_lambda = adanet_lambda
        self._adanet_beta = adanet_beta
        self._use_bias = use_bias
        self._name = name

    @property
    def name(self) -> str:
        return self._name or "complexity_regularized"

    def _init_weight(self, subnetwork: Subnetwork, num_subnetworks: int,
                     logits_dimension: int, device, dtype) -> torch.Tensor:
        """Default init: SCALAR/VECTOR -> 1/N uniform average; MATRIX -> zeros
        (reference weighted.py:400-426)."""
        t = self._mixture_weight_type
        if self._mixture_weight_initializer is not None:
            init = self._mixture_weight_initializer
            if callable(init):
                return init(t, logits_dimension, device=device, dtype=dtype)
            return torch.as_tensor(init, device=device, dtype=dtype).clone()
        if t == MixtureWeightType.SCALAR:
            return torch.full((), 1.0 / num_subnetworks, device=device,
                              dtype=dtype)
        if t == MixtureWeightType.VECTOR:
            return torch.full((logits_dimension,), 1.0 / num_subnetworks,
                              device=device, dtype=dtype)
        if t == MixtureWeightType.MATRIX:
            # Probe last-layer width lazily from the module if available.
            d = getattr(subnetwork.module, "last_layer_dim", None)
            if d is None:
                raise ValueError(
                    "MATRIX mixture weights need subnetwork.module to expose "
                    "`last_layer_dim` (int width of its last layer)")
            return torch.zeros((int(d), logits_dimension), device=device,
                               dtype=dtype)
        raise ValueError("unknown mixture weight type %r" % (t,))

    def build_ensemble(self, subnetworks, previous_ensemble_subnetworks,
                       features, labels, logits_dimension, training,
                       previous_ensemble, device=None) -> ComplexityRegularized:
        """Builds the weighted ensemble (reference weighted.py:253-561).

        ``previous_ensemble_subnetworks`` selects which members of
        ``previous_ensemble`` survive into this candidate (pruning,
        reference weighted.py:269-292); surviving members keep their frozen
        subnetwork modules and — when ``warm_start_mixture_weights`` — their
        learned mixture-weight values.
        """
        device = device or (features.device if isinstance(
            features, torch.Tensor) else "cpu")
        # Mixture weights are small; keep them fp32 for stable L1/optimizer
        # math (the fused mixer kernel reads them as fp32 scalars).
        dtype = torch.float32

        weighted = []
        num_subnetworks = len(subnetworks)
        prev = list(previous_ensemble_subnetworks or [])
        if prev and previous_ensemble is not None:
            num_subnetworks += len(prev)
            for prev_ws in previous_ensemble.weighted_subnetworks:
                if prev_ws.subnetwork not in prev:
                    continue  # pruned (reference weighted.py:276-279)
                if self._warm_start_mixture_weights:
                    w = prev_ws.weight.detach().clone().to(device=device,
                                                           dtype=dtype)
                else:
                    w = self._init_weight(prev_ws.subnetwork, num_subnetworks,
                                          logits_dimension, device, dtype)
                weighted.append(
                    WeightedSubnetwork(
                        prev_ws.subnetwork, w, self._mixture_weight_type,
                        iteration_number=prev_ws.iteration_number,
                        builder_name=prev_ws.builder_name))
        for sub in subnetworks:
            w = self._init_weight(sub, num_subnetworks, logits_dimension,
                                  device, dtype)
            weighted.append(
                WeightedSubnetwork(sub, w, self._mixture_weight_type,
                                   builder_name=sub.name))

        bias = None
        if self._use_bias:
            # Warm-start bias from the previous ensemble (weighted.py:338-349).
            if (self._warm_start_mixture_weights and previous_ensemble
                    is not None and previous_ensemble.bias is not None):
                bias = previous_ensemble.bias.detach().clone().to(
                    device=device, dtype=dtype)
            else:
                bias = torch.zeros((logits_dimension,), device=device,
                                   dtype=dtype)
        ens = ComplexityRegularized(weighted, bias, self._adanet_lambda,
                                    self._adanet_beta, self._use_bias)
        return ens.to(device)

    def build_optimizer(self, ensemble: ComplexityRegularized,
                        iteration: int = 0):
        """Mixture-weight optimizer (reference weighted.py:606-617). None ->
        mixture weights stay at their initializer (reference tf.no_op())."""
        if self._optimizer is None:
            return None
        params = ensemble.mixture_parameters()
        if not params:
            return None
        if callable(self._optimizer) and not hasattr(self._optimizer, "step"):
            return self._optimizer(params)
        return self._optimizer
