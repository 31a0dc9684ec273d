"""HipDropout (K3): stateless counter-based-RNG dropout.

The mask is a pure function of (seed, element index) so backward recomputes
it — no mask tensor is stored or transferred. Seeds advance from a
per-process counter seeded by torch's RNG so runs are reproducible under
torch.manual_seed.
"""

from __future__ import annotations

import torch
from torch import nn

from adanet_amd.ops import _extension

_seed_counter = [0]


def _next_seed() -> int:
    if _seed_counter[0] == 0:
        _seed_counter[0] = int(torch.initial_seed()) & 0x7FFFFFFFFFFF or 1
    _seed_counter[0] += 0x9E3779B9
    return _seed_counter[0]


class _DropoutFn(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, p, seed):
        ext = _extension.require()
        x = x.contiguous()
        y = torch.empty_like(x)
        ext.dropout_fwd(x, y, p, seed)
        ctx.p = p
        ctx.seed = seed
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _extension.require()
        dy = dy.contiguous()
        dx = torch.empty_like(dy)
        ext.dropout_bwd(dy, dx, ctx.p, ctx.seed)
        return dx, None, None


class HipDropout(nn.Module):

    def __init__(self, p: float = 0.5):
        super().__init__()
        if not 0.0 <= p < 1.0:
            raise ValueError("dropout p must be in [0, 1)")
        self.p = p

    def forward(self, x):
        if not self.training or self.p == 0.0:
            return x
        if x.is_cuda:
            return _DropoutFn.apply(x.to(torch.bfloat16), self.p,
                                    _next_seed())
        return torch.nn.functional.dropout(x, self.p, training=True)
