"""ComplexityRegularizedEnsembler — the AdaNet objective (Eq. 4).

MI355X-native re-implementation of reference adanet/ensemble/weighted.py:
the ensemble logits are

    F(x) = b + sum_j mix(w_j, h_j(x))

where ``mix`` is elementwise scale (SCALAR), per-class scale (VECTOR) or
``last_layer @ W_j`` (MATRIX) — reference weighted.py:400-454, 545-561 —
and the complexity regularization term is

    sum_j (lambda * r(h_j) + beta) * ||w_j||_1        (weighted.py:563-604)

MI355X design notes:
  * SCALAR/VECTOR mixture weights live in ONE flat fp32 parameter
    [J] / [J, C] owned by the ensemble (the reference keeps per-subnetwork
    TF variables). The fused mixer kernel (csrc/mixer.hip) reads the flat
    buffer directly, the L1 penalty is three tiny tensor ops instead of a
    per-member chain, and the mixture optimizer is ONE fused update.
    ``WeightedSubnetwork.weight`` stays the per-member view for API parity.
  * MATRIX weights stay per-member parameters (shapes differ per member);
    their mix is a plain library GEMM.
  * Frozen member logits come from the iteration's HBM logit cache; the
    mixer never re-runs frozen subnetworks.
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
    """A subnetwork paired with its mixture weight.

    Reference: adanet/ensemble/weighted.py:43-85. For SCALAR/VECTOR types
    ``weight`` is a view into the owning ensemble's flat parameter; for
    MATRIX it is this module's own parameter.
    """

    def __init__(self, subnetwork: Subnetwork, mixture_weight_type: str,
                 iteration_number: int = 0, builder_name: str = "",
                 matrix_weight: Optional[torch.Tensor] = None):
        super().__init__()
        self.subnetwork = subnetwork
        self.mixture_weight_type = mixture_weight_type
        self.iteration_number = iteration_number
        self.builder_name = builder_name or subnetwork.name
        if matrix_weight is not None:
            self.matrix_weight = nn.Parameter(matrix_weight)
        else:
            self.register_parameter("matrix_weight", None)
        self._flat_getter = None  # set by the owning ensemble

    @property
    def name(self) -> str:
        return self.builder_name

    @property
    def weight(self) -> torch.Tensor:
        if self.matrix_weight is not None:
            return self.matrix_weight
        if self._flat_getter is None:
            raise RuntimeError("WeightedSubnetwork not attached to ensemble")
        return self._flat_getter()

    def mix(self, last_layer: torch.Tensor,
            logits: torch.Tensor) -> torch.Tensor:
        """Weighted logits contribution (reference weighted.py:427-454)."""
        if self.mixture_weight_type == MixtureWeightType.MATRIX:
            # last_layer @ W, supporting rank-3 last layers by flattening
            # the middle dims (reference weighted.py:434-451). Plain library
            # GEMM (fp32 weights).
            ll = last_layer
            if ll.dim() > 2:
                ll = ll.reshape(ll.shape[0], -1)
            return ll.to(self.weight.dtype) @ self.weight
        return logits * self.weight


class ComplexityRegularized(Ensemble):
    """The ensemble produced by :class:`ComplexityRegularizedEnsembler`.

    Reference: adanet/ensemble/weighted.py:88-132.
    """

    def __init__(self, weighted_subnetworks: Sequence[WeightedSubnetwork],
                 bias: Optional[torch.Tensor], adanet_lambda: float,
                 adanet_beta: float, use_bias: bool,
                 mixture_weight_type: str,
                 flat_weights: Optional[torch.Tensor] = None):
        super().__init__()
        self.weighted_subnetworks = nn.ModuleList(weighted_subnetworks)
        self.adanet_lambda = float(adanet_lambda)
        self.adanet_beta = float(adanet_beta)
        self.use_bias = use_bias
        self.mixture_weight_type = mixture_weight_type
        if flat_weights is not None:
            self.mixture_weights = nn.Parameter(flat_weights)
            for j, ws in enumerate(self.weighted_subnetworks):
                ws._flat_getter = (
                    lambda jj=j: self.mixture_weights[jj])
        else:
            self.register_parameter("mixture_weights", None)
        # (lambda * r_j + beta) per member, for the fused L1 penalty.
        coefs = torch.tensor([
            self.adanet_lambda * float(ws.subnetwork.complexity) +
            self.adanet_beta for ws in weighted_subnetworks
        ], dtype=torch.float32)
        self.register_buffer("creg_coefs", coefs)
        if use_bias and bias is not None:
            self.bias = nn.Parameter(bias)
        else:
            self.register_parameter("bias", None)

    @property
    def subnetworks(self) -> List[Subnetwork]:
        return [ws.subnetwork for ws in self.weighted_subnetworks]

    def mixture_parameters(self):
        params = []
        if self.mixture_weights is not None:
            params.append(self.mixture_weights)
        for ws in self.weighted_subnetworks:
            if ws.matrix_weight is not None:
                params.append(ws.matrix_weight)
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

        SCALAR/VECTOR dispatch to the fused weighted-sum HIP kernel (K5)
        reading the flat weight buffer; MATRIX falls back to per-member
        GEMMs."""
        if self.mixture_weights is not None and len(sub_logits) > 0:
            from adanet_amd.ops import mixer
            return mixer.weighted_sum_logits(sub_logits,
                                             self.mixture_weights, self.bias)
        total = None
        for ws, ll, lg in zip(self.weighted_subnetworks, sub_last_layers,
                              sub_logits):
            contrib = ws.mix(ll, lg)
            total = contrib if total is None else total + contrib
        if self.bias is not None:
            total = total + self.bias
        return total

    def complexity_regularization(self) -> torch.Tensor:
        """sum_j (lambda*r_j + beta) * ||w_j||_1 (reference weighted.py:
        563-604). Flat-weight types: three tensor ops total."""
        if self.adanet_lambda == 0.0 and self.adanet_beta == 0.0:
            dev = self.creg_coefs.device
            return torch.zeros((), device=dev)
        if self.mixture_weights is not None:
            w = self.mixture_weights
            l1 = w.abs() if w.dim() == 1 else w.abs().sum(dim=-1)
            return (self.creg_coefs * l1).sum()
        total = None
        for coef, ws in zip(self.creg_coefs, self.weighted_subnetworks):
            term = coef * ws.weight.abs().sum()
            total = term if total is None else total + term
        if total is None:
            total = torch.zeros((), device=self.creg_coefs.device)
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

    def _default_init(self, num_subnetworks: int, logits_dimension: int,
                      device, dtype):
        """SCALAR/VECTOR -> 1/N uniform average (reference weighted.py:
        400-426)."""
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
        raise ValueError("unknown mixture weight type %r" % (t,))

    def _matrix_init(self, subnetwork: Subnetwork, logits_dimension: int,
                     device, dtype):
        if self._mixture_weight_initializer is not None:
            init = self._mixture_weight_initializer
            if callable(init):
                return init(MixtureWeightType.MATRIX, logits_dimension,
                            device=device, dtype=dtype)
            return torch.as_tensor(init, device=device, dtype=dtype).clone()
        d = getattr(subnetwork.module, "last_layer_dim", None)
        if d is None:
            raise ValueError(
                "MATRIX mixture weights need subnetwork.module to expose "
                "`last_layer_dim` (int width of its last layer)")
        # tf.zeros_initializer (reference weighted.py:412-414).
        return torch.zeros((int(d), logits_dimension), device=device,
                           dtype=dtype)

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
        dtype = torch.float32  # mixture weights are fp32 by design
        matrix = self._mixture_weight_type == MixtureWeightType.MATRIX

        weighted: List[WeightedSubnetwork] = []
        init_values: List[torch.Tensor] = []
        num_subnetworks = len(subnetworks)
        prev = list(previous_ensemble_subnetworks or [])
        if prev and previous_ensemble is not None:
            num_subnetworks += len(prev)
            for k, prev_ws in enumerate(
                    previous_ensemble.weighted_subnetworks):
                if prev_ws.subnetwork not in prev:
                    continue  # pruned (reference weighted.py:276-279)
                if matrix:
                    if self._warm_start_mixture_weights:
                        w = prev_ws.weight.detach().clone().to(
                            device=device, dtype=dtype)
                    else:
                        w = self._matrix_init(prev_ws.subnetwork,
                                              logits_dimension, device,
                                              dtype)
                    weighted.append(
                        WeightedSubnetwork(
                            prev_ws.subnetwork, self._mixture_weight_type,
                            iteration_number=prev_ws.iteration_number,
                            builder_name=prev_ws.builder_name,
                            matrix_weight=w))
                else:
                    if self._warm_start_mixture_weights:
                        init_values.append(prev_ws.weight.detach().clone().to(
                            device=device, dtype=dtype))
                    else:
                        init_values.append(
                            self._default_init(num_subnetworks,
                                               logits_dimension, device,
                                               dtype))
                    weighted.append(
                        WeightedSubnetwork(
                            prev_ws.subnetwork, self._mixture_weight_type,
                            iteration_number=prev_ws.iteration_number,
                            builder_name=prev_ws.builder_name))
        for sub in subnetworks:
            if matrix:
                weighted.append(
                    WeightedSubnetwork(sub, self._mixture_weight_type,
                                       builder_name=sub.name,
                                       matrix_weight=self._matrix_init(
                                           sub, logits_dimension, device,
                                           dtype)))
            else:
                init_values.append(
                    self._default_init(num_subnetworks, logits_dimension,
                                       device, dtype))
                weighted.append(
                    WeightedSubnetwork(sub, self._mixture_weight_type,
                                       builder_name=sub.name))

        flat = None
        if not matrix and init_values:
            # SCALAR entries are 0-d; broadcast VECTOR entries to [C].
            if self._mixture_weight_type == MixtureWeightType.VECTOR:
                init_values = [
                    v.expand(logits_dimension).clone() if v.dim() == 0 else v
                    for v in init_values
                ]
            flat = torch.stack([v.reshape(-1) for v in init_values])
            if self._mixture_weight_type == MixtureWeightType.SCALAR:
                flat = flat.reshape(len(init_values))

        bias = None
        if self._use_bias:
            # Warm-start bias from the previous ensemble (weighted.py:338-349).
            if (self._warm_start_mixture_weights and previous_ensemble
                    is not None
                    and getattr(previous_ensemble, "bias", None) is not None):
                bias = previous_ensemble.bias.detach().clone().to(
                    device=device, dtype=dtype)
            else:
                bias = torch.zeros((logits_dimension,), device=device,
                                   dtype=dtype)
        ens = ComplexityRegularized(weighted, bias, self._adanet_lambda,
                                    self._adanet_beta, self._use_bias,
                                    self._mixture_weight_type,
                                    flat_weights=flat)
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
