"""simple_dnn: the canonical AdaNet-paper DNN search space.

Mirrors reference adanet/examples/simple_dnn.py:26-213 — at iteration t the
Generator proposes two candidates: a DNN with the same depth as the
previous best subnetwork and one a layer deeper. complexity r(h) =
sqrt(depth) (reference :90), shared state carries num_layers (:202-213).

MI355X-native: layers are HipLinear (fused MFMA GEMM + bias + ReLU
epilogue) with HipDropout; optimizers are the fused K4 kernels.
"""

from __future__ import annotations

import functools
from typing import List, Optional

import torch
from torch import nn

from adanet_amd.ops.dropout import HipDropout
from adanet_amd.ops.linear import HipLinear
from adanet_amd.ops.optim import FusedSGD
from adanet_amd.subnetwork.generator import (Builder, Generator, Subnetwork,
                                             SubnetworkModule)
from adanet_amd.subnetwork.report import Report


def _feature_dim(features) -> int:
    t = features
    if isinstance(features, dict):
        t = features[sorted(features.keys())[0]]
    return int(torch.tensor(t.shape[1:]).prod())


class _DNNModule(SubnetworkModule):

    def __init__(self, in_dim: int, num_layers: int, layer_size: int,
                 logits_dim: int, dropout: float):
        super().__init__()
        import os
        unfused = os.environ.get("ADANET_UNFUSED_DROPOUT", "") not in ("", "0")
        layers = []
        d = in_dim
        for _ in range(num_layers):
            # dropout is FUSED into the GEMM's relu epilogue (gemm.hip
            # act=3, stateless counter RNG): the paper's relu->dropout
            # layer (reference simple_dnn.py:77-81) costs zero extra
            # kernels and stays hipGraph-capturable.
            layers.append(HipLinear(d, layer_size, activation="relu",
                                    dropout=0.0 if unfused else dropout))
            if unfused and dropout > 0:
                layers.append(HipDropout(dropout))
            d = layer_size
        self.hidden = nn.Sequential(*layers)
        self.logits_layer = HipLinear(d, logits_dim)
        self.last_layer_dim = d

    def forward(self, features):
        x = features
        if isinstance(features, dict):
            x = features[sorted(features.keys())[0]]
        if x.dim() > 2:
            x = x.reshape(x.shape[0], -1)
        last = self.hidden(x) if len(self.hidden) else x
        logits = self.logits_layer(last)
        return last, logits


class _SimpleDNNBuilder(Builder):
    """Reference adanet/examples/simple_dnn.py:26-133."""

    def __init__(self, optimizer_fn, layer_size: int, num_layers: int,
                 learn_mixture_weights: bool, dropout: float, seed=None,
                 name_suffix: str = "", mixture_optimizer_fn=None):
        self._optimizer_fn = optimizer_fn
        self._mixture_optimizer_fn = mixture_optimizer_fn or optimizer_fn
        self._layer_size = layer_size
        self._num_layers = num_layers
        self._learn_mixture_weights = learn_mixture_weights
        self._dropout = dropout
        self._seed = seed
        self._name_suffix = name_suffix

    @property
    def name(self) -> str:
        # Reference naming: "linear" for 0 layers else "<n>_layer_dnn" (:129).
        base = ("linear" if self._num_layers == 0 else
                "{}_layer_dnn".format(self._num_layers))
        return base + self._name_suffix

    def build_subnetwork(self, features, logits_dimension, training,
                         previous_ensemble=None) -> Subnetwork:
        if self._seed is not None:
            # stable across processes (python str hash is randomized)
            suffix_code = sum(ord(c) * (i + 1)
                              for i, c in enumerate(self._name_suffix))
            torch.manual_seed(self._seed + self._num_layers * 7919 +
                              suffix_code)
        module = _DNNModule(_feature_dim(features), self._num_layers,
                            self._layer_size, logits_dimension,
                            self._dropout if training else 0.0)
        # complexity measure: r(h) = sqrt(depth) (reference :90).
        return Subnetwork(module=module,
                          complexity=float(self._num_layers ** 0.5),
                          shared={"num_layers": self._num_layers},
                          name=self.name)

    def build_optimizer(self, params, iteration: int = 0):
        return self._optimizer_fn(params)

    def build_mixture_weights_optimizer(self, params, iteration: int = 0):
        if not self._learn_mixture_weights:
            return None
        params = list(params)
        if not params:
            return None
        return self._mixture_optimizer_fn(params)

    def build_subnetwork_report(self) -> Report:
        return Report(hparams={"layer_size": self._layer_size,
                               "num_layers": self._num_layers},
                      attributes={"complexity": self._num_layers ** 0.5},
                      metrics={})


class Generator(Generator):
    """Two candidates per iteration: same depth + one deeper
    (reference simple_dnn.py:134-213). ``num_restarts`` widens the search
    with independently-initialized copies of each candidate ("_r<i>"
    suffixes) — the pool a round-robin placement spreads over the node's
    GPUs (2*restarts candidates per iteration)."""

    def __init__(self, optimizer_fn=None, layer_size: int = 32,
                 initial_num_layers: int = 0,
                 learn_mixture_weights: bool = False, dropout: float = 0.0,
                 seed: Optional[int] = None, num_restarts: int = 1,
                 mixture_optimizer_fn=None):
        if optimizer_fn is None:
            optimizer_fn = functools.partial(FusedSGD, lr=0.01)
        self._builder_fn = functools.partial(
            _SimpleDNNBuilder,
            optimizer_fn=optimizer_fn,
            mixture_optimizer_fn=mixture_optimizer_fn,
            layer_size=layer_size,
            learn_mixture_weights=learn_mixture_weights,
            dropout=dropout,
            seed=seed)
        self._initial_num_layers = initial_num_layers
        self._num_restarts = max(1, int(num_restarts))

    def generate_candidates(self, previous_ensemble, iteration_number,
                            previous_ensemble_reports, all_reports,
                            config=None) -> List[Builder]:
        num_layers = self._initial_num_layers
        if previous_ensemble is not None:
            last = previous_ensemble.subnetworks[-1]
            shared = last.shared or {}
            num_layers = int(shared.get("num_layers",
                                        self._initial_num_layers))
        out = []
        for r in range(self._num_restarts):
            suffix = "" if r == 0 else "_r{}".format(r)
            out.append(self._builder_fn(num_layers=num_layers,
                                        name_suffix=suffix))
            out.append(self._builder_fn(num_layers=num_layers + 1,
                                        name_suffix=suffix))
        return out
