"""Canned estimators: Linear and DNN (the analog of the tf.estimator canned
estimators the reference's AutoEnsembleEstimator wraps,
adanet/autoensemble/estimator.py:28-220 docs examples).

A canned estimator here is a small spec object exposing
``build_model(features, logits_dimension, training) -> nn.Module`` and
``make_optimizer(params)`` — exactly what _BuilderFromSubestimator needs.
"""

from __future__ import annotations

import functools
from typing import Callable, Optional, Sequence

import torch
from torch import nn

from adanet_amd.ops.dropout import HipDropout
from adanet_amd.ops.linear import HipLinear
from adanet_amd.ops.optim import FusedSGD
from adanet_amd.subnetwork.generator import SubnetworkModule


def _feature_dim(features) -> int:
    t = features
    if isinstance(features, dict):
        t = features[sorted(features.keys())[0]]
    return int(torch.tensor(t.shape[1:]).prod())


class _FlattenInput(nn.Module):

    def forward(self, features):
        x = features
        if isinstance(features, dict):
            x = features[sorted(features.keys())[0]]
        if x.dim() > 2:
            x = x.reshape(x.shape[0], -1)
        return x


class _LinearModule(SubnetworkModule):

    def __init__(self, in_dim, logits_dim):
        super().__init__()
        self.flatten = _FlattenInput()
        self.logits_layer = HipLinear(in_dim, logits_dim)
        self.last_layer_dim = in_dim

    def forward(self, features):
        x = self.flatten(features)
        return x, self.logits_layer(x)


class _DNNModule(SubnetworkModule):

    def __init__(self, in_dim, hidden_units, logits_dim, dropout):
        super().__init__()
        self.flatten = _FlattenInput()
        layers = []
        d = in_dim
        for h in hidden_units:
            layers.append(HipLinear(d, h, activation="relu"))
            if dropout:
                layers.append(HipDropout(dropout))
            d = h
        self.hidden = nn.Sequential(*layers)
        self.logits_layer = HipLinear(d, logits_dim)
        self.last_layer_dim = d

    def forward(self, features):
        x = self.flatten(features)
        last = self.hidden(x)
        return last, self.logits_layer(last)


class LinearEstimator(object):
    """Linear model (analog of tf.estimator.LinearEstimator)."""

    def __init__(self, head=None, optimizer: Optional[Callable] = None,
                 seed: Optional[int] = None):
        self.head = head
        self._optimizer = optimizer or functools.partial(FusedSGD, lr=0.01)
        self._seed = seed

    def build_model(self, features, logits_dimension, training):
        if self._seed is not None:
            torch.manual_seed(self._seed)
        return _LinearModule(_feature_dim(features), logits_dimension)

    def make_optimizer(self, params):
        return self._optimizer(params)


class DNNEstimator(object):
    """Fully-connected DNN (analog of tf.estimator.DNNEstimator)."""

    def __init__(self, head=None, hidden_units: Sequence[int] = (300,),
                 optimizer: Optional[Callable] = None, dropout: float = 0.0,
                 seed: Optional[int] = None):
        self.head = head
        self.hidden_units = list(hidden_units)
        self._optimizer = optimizer or functools.partial(FusedSGD, lr=0.01)
        self.dropout = dropout
        self._seed = seed

    def build_model(self, features, logits_dimension, training):
        if self._seed is not None:
            torch.manual_seed(self._seed)
        return _DNNModule(_feature_dim(features), self.hidden_units,
                          logits_dimension, self.dropout if training else 0.0)

    def make_optimizer(self, params):
        return self._optimizer(params)


class _DNNLinearCombinedModule(SubnetworkModule):
    """Wide & deep: logits = linear(x) + dnn_head(dnn(x)); last layer is
    the deep tower's top (the part downstream ensembles can extend)."""

    def __init__(self, in_dim, hidden_units, logits_dim, dropout):
        super().__init__()
        self.flatten = _FlattenInput()
        self.wide = HipLinear(in_dim, logits_dim)
        layers = []
        d = in_dim
        for h in hidden_units:
            layers.append(HipLinear(d, h, activation="relu"))
            if dropout:
                layers.append(HipDropout(dropout))
            d = h
        self.deep = nn.Sequential(*layers)
        self.deep_logits = HipLinear(d, logits_dim, bias=False)
        self.last_layer_dim = d

    def forward(self, features):
        x = self.flatten(features)
        last = self.deep(x)
        return last, self.wide(x) + self.deep_logits(last)


class DNNLinearCombinedEstimator(object):
    """Wide & deep (analog of tf.estimator.DNNLinearCombinedEstimator —
    the canned family the reference's AutoEnsemble examples pool)."""

    def __init__(self, head=None, hidden_units: Sequence[int] = (300,),
                 optimizer: Optional[Callable] = None,
                 linear_optimizer: Optional[Callable] = None,
                 dropout: float = 0.0, seed: Optional[int] = None):
        self.head = head
        self.hidden_units = list(hidden_units)
        # One fused optimizer over both towers (torch joint training; the
        # reference's separate FTRL/Adagrad split is an async-PS artifact).
        self._optimizer = (optimizer or linear_optimizer
                           or functools.partial(FusedSGD, lr=0.01))
        self.dropout = dropout
        self._seed = seed

    def build_model(self, features, logits_dimension, training):
        if self._seed is not None:
            torch.manual_seed(self._seed)
        return _DNNLinearCombinedModule(
            _feature_dim(features), self.hidden_units, logits_dimension,
            self.dropout if training else 0.0)

    def make_optimizer(self, params):
        return self._optimizer(params)
