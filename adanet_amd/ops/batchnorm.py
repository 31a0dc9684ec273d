"""HipBatchNorm2d (K8): per-channel BatchNorm on native CDNA4 kernels.

fp32 statistics/affine under bf16 NCHW activations, torch-compatible
running-stat semantics (momentum EMA, unbiased running variance). Replaces
the round-1 nn.BatchNorm2d fallback in the NASNet cells (reference BN arg
scopes, research/improve_nas/trainer/nasnet.py:127-233). CPU falls back to
F.batch_norm in fp32; on GPU the extension is required (fail-loud).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import nn

from adanet_amd.ops import _extension


class _BatchNormFn(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, gamma, beta, running_mean, running_var, momentum,
                eps, training):
        ext = _extension.require()
        x = x.contiguous()
        C = x.shape[1]
        y = torch.empty_like(x)
        if training:
            mean = torch.empty((C,), device=x.device, dtype=torch.float32)
            rstd = torch.empty((C,), device=x.device, dtype=torch.float32)
            ext.batchnorm_stats(x, mean, rstd, running_mean, running_var,
                                eps, momentum)
        else:
            mean = running_mean
            rstd = (running_var + eps).rsqrt()
        ext.batchnorm_norm(x, y, mean, rstd, gamma, beta)
        ctx.save_for_backward(x, gamma, mean, rstd)
        ctx.training = training
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _extension.require()
        x, gamma, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        C = x.shape[1]
        dx = torch.empty_like(x)
        sdy = torch.empty((C,), device=x.device, dtype=torch.float32)
        sdyx = torch.empty((C,), device=x.device, dtype=torch.float32)
        if ctx.training:
            ext.batchnorm_bwd(x, dy, dx, mean, rstd, gamma, sdy, sdyx)
        else:
            # eval mode: stats are constants -> dx = dy * g * rstd; the
            # reduce kernel still provides dgamma/dbeta.
            ext.batchnorm_bwd(x, dy, dx, mean, rstd, gamma, sdy, sdyx)
            scale = (gamma if gamma is not None else
                     torch.ones_like(rstd)) * rstd
            dx = (dy.float() * scale.view(1, -1, 1, 1)).to(dy.dtype)
        dgamma = sdyx if gamma is not None else None
        dbeta = sdy
        return dx, dgamma, dbeta, None, None, None, None, None


class HipBatchNorm2d(nn.Module):
    """Drop-in BatchNorm2d: bf16 activations, fp32 stats, native kernels."""

    def __init__(self, num_features: int, momentum: float = 0.1,
                 eps: float = 1e-3, affine: bool = True):
        super().__init__()
        self.num_features = num_features
        self.momentum = momentum
        self.eps = eps
        if affine:
            self.weight = nn.Parameter(
                torch.ones(num_features, dtype=torch.float32))
            self.bias = nn.Parameter(
                torch.zeros(num_features, dtype=torch.float32))
        else:
            self.register_parameter("weight", None)
            self.register_parameter("bias", None)
        self.register_buffer(
            "running_mean", torch.zeros(num_features, dtype=torch.float32))
        self.register_buffer(
            "running_var", torch.ones(num_features, dtype=torch.float32))
        self.register_buffer(
            "num_batches_tracked", torch.zeros((), dtype=torch.long))

    def forward(self, x):
        # dtype pinning: stats/affine are fp32 by design; a blanket
        # module.to(bf16) without restore_fp32_params must not break BN.
        if self.running_mean.dtype != torch.float32:
            self.running_mean = self.running_mean.float()
            self.running_var = self.running_var.float()
        if self.weight is not None and self.weight.dtype != torch.float32:
            self.weight.data = self.weight.data.float()
            self.bias.data = self.bias.data.float()
        if x.is_cuda:
            if self.training:
                self.num_batches_tracked += 1
            return _BatchNormFn.apply(
                x.to(torch.bfloat16), self.weight, self.bias,
                self.running_mean, self.running_var, self.momentum, self.eps,
                self.training)
        out = F.batch_norm(
            x.float(), self.running_mean, self.running_var,
            self.weight.float() if self.weight is not None else None,
            self.bias.float() if self.bias is not None else None,
            self.training, self.momentum, self.eps)
        return out.to(x.dtype)

    def extra_repr(self):
        return "features=%d, eps=%g, momentum=%g" % (
            self.num_features, self.eps, self.momentum)
