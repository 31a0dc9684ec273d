"""HipLinear: bf16 dense layer on the fused MFMA GEMM (K1).

Forward is one gemm_nt_bf16 call with the bias+ReLU epilogue fused into the
GEMM (adanet_amd/csrc/gemm.hip). Backward is TRANSPOSE-FREE: the
transposed-staging GEMMs (csrc/gemm_tn.hip, ds_read_b64_tr_b16) consume
dz/W/x in their stored K-major layouts, and under `direct_grad_writes`
their epilogues land dW/db straight in the optimizer's arena grad views
(accumulate act=2, or plain overwrite when the optimizer skipped the
arena memset for single-write params):

    dz = dReLU(dY)              (fused with the db colsum when direct)
    dX = dz @ W                 (trans_b staging; per-shape vs composite)
    dW = dz^T @ X               (tt staging -> arena view)
    db = colsum(dz)             (-> arena view)

The ReLU mask is recovered from the saved forward OUTPUT (y > 0), so the
fused epilogue never materializes a mask tensor.

Out-features are padded up to a multiple of 32 internally so every GEMM and
logit buffer keeps a 16 B-aligned row stride (global_load_lds fast path) AND
stays a valid K%32 reduction dim for the backward dX GEMM; callers see the
narrow [B, N] view.

Reference dependency being replaced: tf.layers.dense in subnetworks
(adanet/examples/simple_dnn.py:74-86).
"""

from __future__ import annotations

import math
from typing import Optional

import torch
from torch import nn

from adanet_amd.ops import _extension


_DIRECT_GRAD = False


class direct_grad_writes:
    """Enable direct-to-arena gradient writes inside this context.

    With this on, _LinearFn.backward writes dW/db straight into the
    parameter's existing `.grad` tensor using the GEMM accumulate epilogue
    (act=2) / colsum accum mode, and returns None to autograd — eliminating
    AccumulateGrad's separate add kernel and the colsum pre-zero fill
    (measured 5-6% of step GPU time, profiles/bench_kernel_stats_r01d.txt).

    Semantics match autograd exactly (add into existing grad), so it is only
    valid when gradients are consumed via `.grad` (optimizer steps), not via
    `torch.autograd.grad`. The training engine wraps its backward pass in
    this context; arena-pinned grad views (ops/optim.py) make the target
    addresses stable across hipGraph replays.
    """

    def __enter__(self):
        global _DIRECT_GRAD
        self._prev = _DIRECT_GRAD
        _DIRECT_GRAD = True
        return self

    def __exit__(self, *exc):
        global _DIRECT_GRAD
        _DIRECT_GRAD = self._prev
        return False


def _pad8(n: int) -> int:
    # Pad out-features to a multiple of 32: keeps the row stride 16B-aligned
    # for global_load_lds AND keeps N usable as the reduction dim of the
    # backward dX GEMM (the MFMA fast path needs K % 32 == 0 — a 16-wide
    # padded head sent dX to the generic VALU kernel, 7% of step time).
    return (n + 31) // 32 * 32


def restore_fp32_params(module) -> None:
    """Re-float the deliberately-fp32 leaves after a blanket .to(bf16).

    HipLinear biases and HipLayerNorm affines are fp32 by design (they feed
    fp32 epilogues/statistics); `module.to(torch.bfloat16)` downcasts them,
    so call this afterwards.
    """
    from adanet_amd.ops.conv import HipConv1x1, HipConvNxN
    from adanet_amd.ops.layernorm import HipLayerNorm
    for m in module.modules():
        if isinstance(m, (HipLinear, HipConv1x1, HipConvNxN)) \
                and m.bias is not None:
            m.bias.data = m.bias.data.float()
        if isinstance(m, HipLayerNorm) and m.weight is not None:
            m.weight.data = m.weight.data.float()
            m.bias.data = m.bias.data.float()
        from adanet_amd.ops.batchnorm import HipBatchNorm2d
        if isinstance(m, (torch.nn.modules.batchnorm._BatchNorm,
                          HipBatchNorm2d)):
            # BN statistics/affine stay fp32 under bf16 activations
            # (models/nasnet.py BNfp32 computes in fp32).
            if m.weight is not None:
                m.weight.data = m.weight.data.float()
                m.bias.data = m.bias.data.float()
            if m.running_mean is not None:
                m.running_mean = m.running_mean.float()
                m.running_var = m.running_var.float()


def gemm_nt(a: torch.Tensor, b: torch.Tensor,
            bias: Optional[torch.Tensor] = None,
            activation: Optional[str] = None,
            dropout_p: float = 0.0,
            seed: Optional[torch.Tensor] = None) -> torch.Tensor:
    """C[M,N] = A[M,K] @ B[N,K]^T (+bias, +relu[, +fused dropout]).
    bf16 on GPU, fp32 on CPU."""
    if a.is_cuda:
        ext = _extension.require()
        out = torch.empty((a.shape[0], b.shape[0]), device=a.device,
                          dtype=torch.bfloat16)
        if dropout_p > 0.0 and activation == "relu":
            ext.gemm_nt_bf16(a, b, out, bias, 3, dropout_p, seed)
        else:
            ext.gemm_nt_bf16(a, b, out, bias,
                             1 if activation == "relu" else 0)
        return out
    out = a.float() @ b.float().t()
    if bias is not None:
        out = out + bias.float()
    if activation == "relu":
        out = torch.relu(out)
    return out.to(a.dtype)


def transpose2d(x: torch.Tensor) -> torch.Tensor:
    if x.is_cuda:
        ext = _extension.require()
        out = torch.empty((x.shape[1], x.shape[0]), device=x.device,
                          dtype=x.dtype)
        ext.transpose_bf16(x, out)
        return out
    return x.t().contiguous()


class _LinearFn(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, weight, bias, activation, dropout_p=0.0, seed=None):
        x = x.contiguous()
        y = gemm_nt(x, weight, bias, activation, dropout_p, seed)
        ctx.activation = activation
        ctx.dropout_p = dropout_p
        ctx.has_bias = bias is not None
        # Keep the live Parameter objects so backward can reach their .grad
        # views for direct-to-arena writes (saved_tensors unwraps to plain
        # tensors that drop the identity we need).
        ctx.param_refs = (weight if isinstance(weight, nn.Parameter) else None,
                          bias if isinstance(bias, nn.Parameter) else None)
        ctx.save_for_backward(x, weight, y if activation == "relu" else None)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, y = ctx.saved_tensors
        dy = dy.contiguous()
        need_dx = ctx.needs_input_grad[0]
        need_dw = ctx.needs_input_grad[1]
        if dy.is_cuda:
            ext = _extension.require()
            wparam, bparam = ctx.param_refs
            bg = bparam.grad if (_DIRECT_GRAD and bparam is not None) \
                else None
            direct_b = (ctx.has_bias and bg is not None and bg.is_cuda
                        and bg.dtype == torch.float32
                        and bg.shape == (dy.shape[1],)
                        and bg.is_contiguous())
            db_done = False
            if ctx.activation == "relu":
                # Under the fused relu+dropout epilogue, y>0 identifies
                # kept-and-positive elements: dropout backward is the same
                # mask with a 1/(1-p) scale — no mask recompute needed.
                scale = (1.0 / (1.0 - ctx.dropout_p)
                         if ctx.dropout_p > 0.0 else 1.0)
                dz = torch.empty_like(dy)
                if direct_b and dy.is_contiguous() and y.is_contiguous():
                    # Fused mask + bias-grad colsum: one read of dY/Y
                    # produces dz AND accumulates db into the arena view.
                    ext.relu_bwd_colsum(dy, y, dz, bg, scale)
                    db_done = True
                else:
                    ext.relu_bwd(dy, y, dz, scale)
            else:
                dz = dy
            dx = dw = None
            B, N = dz.shape
            K = weight.shape[1]
            if need_dx:
                # dX = dz @ W: the transposed-staging GEMM consumes W
                # K-major directly (ds_read_b64_tr_b16 fragments) — no
                # W^T materialization. K_r = N is always 32-padded.
                # Measured (profiles/tr_vs_composite_r01.json): tr wins
                # unless the reduction dim STRICTLY dominates both output
                # dims (K_r=4096 > M,N=2048 → 0.87x); at square shapes tr
                # still wins (tr_variants_r01.json: 427 vs ~390 effective
                # at 2048^3), so fall back to transpose+NT only beyond.
                if (weight.stride(0) % 8 == 0 and N % 32 == 0
                        and N <= max(B, K)):
                    dx = torch.empty((B, K), device=dz.device,
                                     dtype=torch.bfloat16)
                    ext.gemm_tr_bf16(dz, weight, dx, None, 0, 0, 1)
                else:
                    wt = transpose2d(weight)  # [K, N]
                    dx = gemm_nt(dz, wt)      # [B, K]
            if need_dw:
                # dW = dz^T @ X: both operands K_r(=batch)-major -> both
                # transposed-staged when the batch is 32-aligned.
                # Direct mode: accumulate (act=2) into the arena grad view
                # and return None — no AccumulateGrad add kernel.
                wg = wparam.grad if (_DIRECT_GRAD and wparam is not None) \
                    else None
                direct_w = (wg is not None and wg.is_cuda
                            and wg.dtype == torch.bfloat16
                            and wg.shape == (N, K) and wg.is_contiguous())
                dw_out = wg if direct_w else torch.empty(
                    (N, K), device=dz.device, dtype=torch.bfloat16)
                # overwrite epilogue when the optimizer skipped the arena
                # zero (single-write params, ops/optim.py zero_grad) — but
                # only on the FIRST dW write this step: a module invoked
                # twice per step (weight sharing in a custom Builder) must
                # accumulate its second write or it would silently clobber
                # the first. The counter resets in zero_grad.
                if direct_w and getattr(wparam, "_adanet_grad_overwrite",
                                        False):
                    writes = getattr(wparam, "_adanet_bwd_writes", 0)
                    wparam._adanet_bwd_writes = writes + 1
                    epi = 0 if writes == 0 else 2
                else:
                    epi = 0 if not direct_w else 2
                if (B % 32 == 0 and dz.stride(0) % 8 == 0
                        and x.stride(0) % 8 == 0):
                    ext.gemm_tr_bf16(dz, x, dw_out, None, epi, 1, 1)
                else:
                    dzt = transpose2d(dz)     # [N, B]
                    xt = transpose2d(x)       # [K, B]
                    ext.gemm_nt_bf16(dzt, xt, dw_out, None, epi)
                dw = None if direct_w else dw_out
            db = None
            if ctx.has_bias and not db_done:
                if direct_b:
                    ext.colsum_bf16(dz, bg, 1)
                else:
                    db = torch.empty((dz.shape[1],), device=dz.device,
                                     dtype=torch.float32)
                    ext.colsum_bf16(dz, db)
        else:
            dzf = dy.float()
            if ctx.activation == "relu":
                dzf = dzf * (y > 0).float() if y is not None else dzf
                if ctx.dropout_p > 0.0:
                    dzf = dzf / (1.0 - ctx.dropout_p)
            dx = (dzf @ weight.float()).to(x.dtype) if need_dx else None
            dw = (dzf.t() @ x.float()).to(weight.dtype) if need_dw else None
            db = dzf.sum(dim=0) if ctx.has_bias else None
        return dx, dw, db, None, None, None


class HipLinear(nn.Module):
    """bf16 Linear with fused bias+ReLU epilogue and padded out-features."""

    def __init__(self, in_features: int, out_features: int, bias: bool = True,
                 activation: Optional[str] = None, device=None,
                 dtype: torch.dtype = torch.bfloat16,
                 dropout: float = 0.0):
        super().__init__()
        if activation not in (None, "relu"):
            raise ValueError("activation must be None or 'relu'")
        if dropout and activation != "relu":
            raise ValueError("fused dropout requires activation='relu'")
        if not 0.0 <= dropout < 1.0:
            raise ValueError("dropout must be in [0, 1)")
        self.in_features = in_features
        self.out_features = out_features
        self.padded_out = _pad8(out_features)
        self.activation = activation
        # Fused dropout (paper search space: relu->dropout per layer,
        # reference simple_dnn.py:77-81). Device counter seed so hipGraph
        # replays draw fresh masks (same scheme as HipDropout).
        self.dropout = float(dropout)
        self.register_buffer("_drop_seed",
                             torch.zeros(1, dtype=torch.int64),
                             persistent=False)
        # persistent snapshot target: the per-step seed snapshot must NOT
        # allocate inside a hipGraph capture (an alloc+free on a candidate
        # stream mid-capture perturbs the graph pool's cross-stream block
        # reuse -- measured as the ensemble mixer reading garbage on
        # replay, benchmarks/nan_inspect2 r02).
        self.register_buffer("_drop_snap",
                             torch.zeros(1, dtype=torch.int64),
                             persistent=False)
        self._drop_seeded = False
        self.weight = nn.Parameter(
            torch.empty((self.padded_out, in_features), device=device,
                        dtype=dtype))
        # exactly one dW write per backward (module used once per step in
        # every shipped builder) — lets the optimizer skip the arena grad
        # memset and the backward GEMM use the overwrite epilogue.
        self.weight._adanet_single_write = True
        if bias:
            self.bias = nn.Parameter(
                torch.zeros((self.padded_out,), device=device,
                            dtype=torch.float32))
        else:
            self.register_parameter("bias", None)
        self.reset_parameters()

    def reset_parameters(self):
        with torch.no_grad():
            # Kaiming-uniform (torch.nn.Linear default), initialized
            # directly on the parameter's device/dtype (no CPU round-trip).
            bound = 1.0 / math.sqrt(self.in_features)
            self.weight.uniform_(-bound, bound)
            if self.padded_out != self.out_features:
                self.weight[self.out_features:].zero_()
            if self.bias is not None:
                self.bias.zero_()

    def forward(self, x):
        if x.dim() > 2:
            x = x.reshape(x.shape[0], -1)
        p = self.dropout if (self.training and self.dropout > 0.0) else 0.0
        if p > 0.0 and x.is_cuda:
            if not self._drop_seeded or self._drop_seed.device != x.device:
                base = int(torch.initial_seed()) & 0x7FFFFFFFFFFF
                self._drop_seed = torch.tensor(
                    [base + id(self) % 100003], dtype=torch.int64,
                    device=x.device)
                self._drop_snap = torch.zeros_like(self._drop_seed)
                self._drop_seeded = True
            # copy_ into the PERSISTENT snapshot buffer: pure device ops,
            # zero allocations inside any capture.
            self._drop_snap.copy_(self._drop_seed)
            self._drop_seed.add_(0x9E3779B9)
            snapshot = self._drop_snap
            y = _LinearFn.apply(x.to(self.weight.dtype), self.weight,
                                self.bias, self.activation, p, snapshot)
        else:
            y = _LinearFn.apply(x.to(self.weight.dtype), self.weight,
                                self.bias, self.activation)
            if p > 0.0:  # CPU fallback: unfused functional dropout
                y = torch.nn.functional.dropout(y, p, training=True)
        if self.padded_out != self.out_features:
            y = y[:, :self.out_features]
        return y

    def extra_repr(self):
        return "in=%d, out=%d (padded %d), act=%s" % (
            self.in_features, self.out_features, self.padded_out,
            self.activation)
