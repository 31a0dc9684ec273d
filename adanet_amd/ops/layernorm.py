"""HipLayerNorm (K8): wave-per-row fused LayerNorm with fp32 stats."""

from __future__ import annotations

import torch
import torch.nn.functional as F
from torch import nn

from adanet_amd.ops import _extension


class _LayerNormFn(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, gamma, beta, eps):
        ext = _extension.require()
        x = x.contiguous()
        B, D = x.shape
        y = torch.empty_like(x)
        mean = torch.empty((B,), device=x.device, dtype=torch.float32)
        rstd = torch.empty((B,), device=x.device, dtype=torch.float32)
        ext.layernorm_fwd(x, gamma, beta, y, mean, rstd, eps)
        ctx.save_for_backward(x, gamma, mean, rstd)
        ctx.has_beta = beta is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _extension.require()
        x, gamma, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        dx = torch.empty_like(x)
        dgamma = dbeta = None
        if gamma is not None:
            dgamma = torch.zeros((x.shape[1],), device=x.device,
                                 dtype=torch.float32)
        if ctx.has_beta:
            dbeta = torch.zeros((x.shape[1],), device=x.device,
                                dtype=torch.float32)
        ext.layernorm_bwd(x, dy, gamma, mean, rstd, dx, dgamma, dbeta)
        return dx, dgamma, dbeta, None


class HipLayerNorm(nn.Module):

    def __init__(self, dim: int, eps: float = 1e-5, affine: bool = True):
        super().__init__()
        self.dim = dim
        self.eps = eps
        if affine:
            self.weight = nn.Parameter(torch.ones(dim, dtype=torch.float32))
            self.bias = nn.Parameter(torch.zeros(dim, dtype=torch.float32))
        else:
            self.register_parameter("weight", None)
            self.register_parameter("bias", None)

    def forward(self, x):
        if x.is_cuda:
            return _LayerNormFn.apply(x.to(torch.bfloat16), self.weight,
                                      self.bias, self.eps)
        xf = x.float()
        out = F.layer_norm(xf, (self.dim,),
                           self.weight.float() if self.weight is not None else None,
                           self.bias.float() if self.bias is not None else None,
                           self.eps)
        return out.to(x.dtype)
