"""HipDropout (K3): stateless counter-based-RNG dropout.

The mask is a pure function of (seed, element index) so backward recomputes
it — no mask tensor is stored. The seed is a DEVICE counter snapshot: the
module's counter advances on-device each forward, which keeps the op
hipGraph-capturable with a fresh mask per replay (graph-eligibility no
longer excludes dropout-bearing search spaces).
"""

from __future__ import annotations

import torch
from torch import nn

from adanet_amd.ops import _extension


class _DropoutFn(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, p, seed_snapshot):
        ext = _extension.require()
        x = x.contiguous()
        y = torch.empty_like(x)
        ext.dropout_fwd(x, y, p, seed_snapshot)
        ctx.p = p
        ctx.save_for_backward(seed_snapshot)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _extension.require()
        (seed_snapshot,) = ctx.saved_tensors
        dy = dy.contiguous()
        dx = torch.empty_like(dy)
        ext.dropout_bwd(dy, dx, ctx.p, seed_snapshot)
        return dx, None, None


class HipDropout(nn.Module):

    def __init__(self, p: float = 0.5):
        super().__init__()
        if not 0.0 <= p < 1.0:
            raise ValueError("dropout p must be in [0, 1)")
        self.p = p
        # Device counter (int64 buffer): advanced on-device each forward so
        # graph replays draw fresh masks. Seeded lazily from torch's RNG.
        self.register_buffer("_seed", torch.zeros(1, dtype=torch.int64),
                             persistent=False)
        # persistent snapshot buffer: no alloc/free inside graph capture
        # (see ops/linear.py HipLinear._drop_snap).
        self.register_buffer("_snap", torch.zeros(1, dtype=torch.int64),
                             persistent=False)
        self._seeded = False

    def forward(self, x):
        if not self.training or self.p == 0.0:
            return x
        if x.is_cuda:
            if not self._seeded or self._seed.device != x.device:
                base = int(torch.initial_seed()) & 0x7FFFFFFFFFFF
                self._seed = torch.tensor(
                    [base + id(self) % 100003], dtype=torch.int64,
                    device=x.device)
                self._snap = torch.zeros_like(self._seed)
                self._seeded = True
            if torch.cuda.is_current_stream_capturing():
                # No alloc/free inside capture (graph-pool cross-stream
                # reuse race, see ops/linear.py); single forward per
                # captured step is the engine's contract.
                self._snap.copy_(self._seed)
                snapshot = self._snap
            else:
                # eager: a fresh clone so backward sees ITS forward's seed
                # even when the module runs twice before backward.
                snapshot = self._seed.clone()
            self._seed.add_(0x9E3779B9)
            return _DropoutFn.apply(x.to(torch.bfloat16), self.p, snapshot)
        return torch.nn.functional.dropout(x, self.p, training=True)
