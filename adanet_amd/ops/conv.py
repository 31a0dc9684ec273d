"""Native pointwise (1x1) convolution on the batched transposed-staging GEMM.

K9: the NASNet space's FLOPs are dominated by the 1x1 pointwise convs inside
separable convs and the ReLU-Conv-BN adaptors (reference
research/improve_nas/trainer/nasnet_utils.py:182 _stacked_separable_conv,
nasnet.py ReluConvBN usage). On MI355X a pointwise conv IS a batched GEMM:

    y_b[Co, HW] = W[Co, Ci] @ x_b[Ci, HW]          (forward: trans_b staging)
    dX_b        = W^T @ dz_b                       (tt staging)
    dW          = sum_b dz_b @ x_b^T               (NT over [C, B*HW] views)

so csrc/gemm_tn.hip's gemm_tr_batched (grid.z = image) runs it on MFMA with
zero im2col and zero transpose copies for fwd/dX. The fast path needs the
GEMM alignment contract (Ci % 32 == 0, HW % 8 == 0); `conv1x1` falls back
to torch's conv (MIOpen) otherwise, so arbitrary channel counts still work.
"""

from __future__ import annotations

from typing import Optional

import torch
import torch.nn.functional as F
from torch import nn

from adanet_amd.ops import _extension


def _native_enabled() -> bool:
    import os
    return os.environ.get("ADANET_NATIVE_CONV", "1") != "0"


def _aligned(x: torch.Tensor, co: int, ci: int) -> bool:
    if not x.is_cuda or x.dtype != torch.bfloat16 or not _native_enabled():
        return False
    hw = x.shape[2] * x.shape[3]
    return ci % 32 == 0 and co % 32 == 0 and hw % 8 == 0 and x.is_contiguous()


class _Conv1x1Fn(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, weight, bias):
        # x [B, Ci, H, W] contiguous; weight [Co, Ci] bf16; bias fp32 [Co].
        ext = _extension.require()
        B, Ci, H, W = x.shape
        Co = weight.shape[0]
        x3 = x.reshape(B, Ci, H * W)
        y = torch.empty((B, Co, H * W), device=x.device, dtype=torch.bfloat16)
        # bias is per-OUTPUT-CHANNEL = per GEMM ROW; the GEMM epilogue's bias
        # is per-column, so add the bias afterwards (broadcast add).
        ext.gemm_tr_batched(weight, x3, y, None, 0, 0, 1)
        if bias is not None:
            y = y + bias.to(torch.bfloat16).reshape(1, Co, 1)
        ctx.save_for_backward(x, weight)
        ctx.has_bias = bias is not None
        return y.reshape(B, Co, H, W)

    @staticmethod
    def backward(ctx, dy):
        ext = _extension.require()
        x, weight = ctx.saved_tensors
        B, Ci, H, W = x.shape
        Co = weight.shape[0]
        dy3 = dy.contiguous().reshape(B, Co, H * W)
        dx = dw = db = None
        if ctx.needs_input_grad[0]:
            dx = torch.empty((B, Ci, H * W), device=x.device,
                             dtype=torch.bfloat16)
            # dX_b = W^T @ dz_b: both operands K(=Co)-major -> tt staging.
            ext.gemm_tr_batched(weight, dy3, dx, None, 0, 1, 1)
            dx = dx.reshape(B, Ci, H, W)
        if ctx.needs_input_grad[1]:
            # dW[Co, Ci] = dz_all[Co, B*HW] @ x_all[Ci, B*HW]^T — channel-
            # major views need one permute copy each (conv is off the
            # headline hot path; the GEMM itself runs on MFMA).
            dz_flat = dy3.permute(1, 0, 2).reshape(Co, B * H * W).contiguous()
            x_flat = x.reshape(B, Ci, H * W).permute(1, 0, 2).reshape(
                Ci, B * H * W).contiguous()
            dw = torch.empty((Co, Ci), device=x.device, dtype=torch.bfloat16)
            ext.gemm_nt_bf16(dz_flat, x_flat, dw, None, 0)
        if ctx.has_bias:
            db = dy3.sum(dim=(0, 2)).float()
        return dx, dw, db


class HipConv1x1(nn.Module):
    """1x1 Conv2d on the batched MFMA GEMM with torch (MIOpen) fallback.

    Drop-in for ``nn.Conv2d(c_in, c_out, 1, bias=...)`` (stride 1). The
    weight keeps Conv2d's [Co, Ci, 1, 1] layout so state dicts interop.
    """

    def __init__(self, in_channels: int, out_channels: int, bias: bool = True,
                 device=None, dtype=None):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        ref = nn.Conv2d(in_channels, out_channels, 1, bias=bias)
        self.weight = nn.Parameter(ref.weight.detach().to(device=device,
                                                          dtype=dtype))
        if bias:
            self.bias = nn.Parameter(ref.bias.detach().float().to(device))
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        w2 = self.weight.reshape(self.out_channels, self.in_channels)
        if _aligned(x, self.out_channels, self.in_channels) \
                and w2.dtype == torch.bfloat16 \
                and w2.stride(0) % 8 == 0 and w2.is_contiguous():
            return _Conv1x1Fn.apply(x, w2, self.bias)
        return torch.nn.functional.conv2d(
            x, self.weight.to(x.dtype),
            self.bias.to(x.dtype) if self.bias is not None else None)

    def extra_repr(self):
        return "in=%d, out=%d (native when C%%32==0)" % (
            self.in_channels, self.out_channels)


def conv1x1(x: torch.Tensor, weight: torch.Tensor,
            bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Functional 1x1 conv: weight [Co, Ci] (or [Co, Ci, 1, 1])."""
    if weight.dim() == 4:
        weight = weight.reshape(weight.shape[0], weight.shape[1])
    co, ci = weight.shape
    if _aligned(x, co, ci) and weight.dtype == torch.bfloat16 \
            and weight.is_contiguous():
        return _Conv1x1Fn.apply(x, weight, bias)
    return torch.nn.functional.conv2d(
        x, weight.reshape(co, ci, 1, 1).to(x.dtype),
        bias.to(x.dtype) if bias is not None else None)


class _DepthwiseFn(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, weight, stride, pad):
        ext = _extension.require()
        B, C, H, W = x.shape
        KS = weight.shape[-1]
        OH = (H + 2 * pad - KS) // stride + 1
        OW = (W + 2 * pad - KS) // stride + 1
        y = torch.empty((B, C, OH, OW), device=x.device, dtype=torch.bfloat16)
        ext.depthwise_fwd(x, weight, y, stride, pad)
        ctx.save_for_backward(x, weight)
        ctx.stride, ctx.pad = stride, pad
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _extension.require()
        x, weight = ctx.saved_tensors
        dy = dy.contiguous()
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dx = torch.empty_like(x)
            ext.depthwise_bwd_dx(dy, weight, dx, ctx.stride, ctx.pad)
        if ctx.needs_input_grad[1]:
            dwf = torch.empty(weight.numel(), device=x.device,
                              dtype=torch.float32)
            ext.depthwise_bwd_dw(x, dy, dwf, ctx.stride, ctx.pad)
            dw = dwf.reshape(weight.shape).to(weight.dtype)
        return dx, dw, None, None


class HipDepthwiseConv2d(nn.Module):
    """Depthwise KxK conv on the native kernel (any channel count),
    drop-in for ``nn.Conv2d(C, C, K, stride, padding, groups=C, bias=False)``.
    Falls back to grouped conv2d (MIOpen) off-GPU or for non-bf16 input.
    """

    def __init__(self, channels: int, kernel_size: int, stride: int = 1,
                 padding: int = 0):
        super().__init__()
        self.channels = channels
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding
        ref = nn.Conv2d(channels, channels, kernel_size, stride=stride,
                        padding=padding, groups=channels, bias=False)
        self.weight = nn.Parameter(ref.weight.detach())  # [C,1,K,K]

    def forward(self, x):
        if (x.is_cuda and x.dtype == torch.bfloat16 and x.is_contiguous()
                and self.weight.dtype == torch.bfloat16
                and _native_enabled()):
            return _DepthwiseFn.apply(x, self.weight.contiguous(),
                                      self.stride, self.padding)
        return torch.nn.functional.conv2d(
            x, self.weight.to(x.dtype), None, self.stride, self.padding,
            groups=self.channels)

    def extra_repr(self):
        return "C=%d, k=%d, s=%d, p=%d (native bf16 path)" % (
            self.channels, self.kernel_size, self.stride, self.padding)


class _Im2colFn(torch.autograd.Function):
    """Batched im2col (one launch for the whole batch, zero-padded K dim)
    with the exact col2im adjoint for backward (csrc/im2col.hip)."""

    @staticmethod
    def forward(ctx, x, K, stride, pad, ckk_pad, oh, ow):
        ext = _extension.require()
        B = x.shape[0]
        out = torch.empty((B, ckk_pad, oh, ow), device=x.device,
                          dtype=x.dtype)
        ext.im2col_bf16(x, out, K, stride, pad)
        ctx.meta = (x.shape, K, stride, pad)
        return out

    @staticmethod
    def backward(ctx, du):
        ext = _extension.require()
        shape, K, stride, pad = ctx.meta
        dx = torch.empty(shape, device=du.device, dtype=du.dtype)
        ext.col2im_bf16(du.contiguous(), dx, K, stride, pad)
        return dx, None, None, None, None, None, None


class HipConvNxN(nn.Module):
    """Dense KxK conv as unfold + the batched MFMA GEMM (K9 tail).

    MIOpen falls back to a naive weight-gradient kernel for these NCHW
    bf16 shapes (6.9 ms/call — 35% of the NASNet step,
    profiles/conv_bench_r01.txt); expressing the conv as im2col (one
    torch.unfold copy, a few MB) + gemm_tr_batched keeps fwd/bwd on the
    MFMA path. The im2col K dim (Ci*K*K) is zero-padded to 32 inside the
    padded weight, so ANY in-channel count works; requires Co % 32 == 0
    (else torch fallback). Drop-in for
    ``nn.Conv2d(ci, co, K, stride, padding, bias=...)``.
    """

    def __init__(self, in_channels: int, out_channels: int, kernel_size: int,
                 stride: int = 1, padding: int = 0, bias: bool = True):
        super().__init__()
        self.in_channels = in_channels
        self.out_channels = out_channels
        self.kernel_size = kernel_size
        self.stride = stride
        self.padding = padding
        ref = nn.Conv2d(in_channels, out_channels, kernel_size,
                        stride=stride, padding=padding, bias=bias)
        self.weight = nn.Parameter(ref.weight.detach())  # [Co, Ci, K, K]
        if bias:
            self.bias = nn.Parameter(ref.bias.detach().float())
        else:
            self.register_parameter("bias", None)

    def forward(self, x):
        K = self.kernel_size
        ckk = self.in_channels * K * K
        ckk_pad = (ckk + 31) // 32 * 32
        if (x.is_cuda and x.dtype == torch.bfloat16 and _native_enabled()
                and self.weight.dtype == torch.bfloat16
                and self.out_channels % 32 == 0):
            B = x.shape[0]
            oh = (x.shape[2] + 2 * self.padding - K) // self.stride + 1
            ow = (x.shape[3] + 2 * self.padding - K) // self.stride + 1
            # batched im2col with the 32-alignment zero-pad folded in
            # (torch's F.unfold is one im2col launch PER IMAGE).
            u = _Im2colFn.apply(x.contiguous(), K, self.stride,
                                self.padding, ckk_pad, oh, ow)
            w2 = self.weight.reshape(self.out_channels, ckk)
            if ckk_pad != ckk:
                w2 = torch.nn.functional.pad(w2, (0, ckk_pad - ckk))
            y = _Conv1x1Fn.apply(u, w2.contiguous(), self.bias)
            return y
        return torch.nn.functional.conv2d(
            x, self.weight.to(x.dtype),
            self.bias.to(x.dtype) if self.bias is not None else None,
            self.stride, self.padding)

    def extra_repr(self):
        return "in=%d, out=%d, k=%d, s=%d (im2col+MFMA when Co%%32==0)" % (
            self.in_channels, self.out_channels, self.kernel_size,
            self.stride)


class _Pool3Fn(torch.autograd.Function):
    """3x3 pooling on the native kernel (csrc/pool.hip): avg is
    count_include_pad=False, max saves the window argmax for an exact
    deterministic gather backward."""

    @staticmethod
    def forward(ctx, x, stride, is_max):
        ext = _extension.require()
        x = x.contiguous()
        N, C, H, W = x.shape
        OH = (H + 2 - 3) // stride + 1
        OW = (W + 2 - 3) // stride + 1
        y = torch.empty((N, C, OH, OW), device=x.device, dtype=x.dtype)
        argmax = None
        if is_max:
            argmax = torch.empty((N, C, OH, OW), device=x.device,
                                 dtype=torch.uint8)
        ext.pool3_fwd(x, y, argmax, stride, int(is_max))
        ctx.save_for_backward(argmax) if argmax is not None else \
            ctx.save_for_backward()
        ctx.meta = (x.shape, stride, is_max)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = _extension.require()
        shape, stride, is_max = ctx.meta
        argmax = ctx.saved_tensors[0] if ctx.saved_tensors else None
        dy = dy.contiguous()
        dx = torch.empty(shape, device=dy.device, dtype=dy.dtype)
        ext.pool3_bwd(dy, argmax, dx, stride, int(is_max))
        return dx, None, None


class HipPool2d(nn.Module):
    """3x3 avg/max pool, stride 1 or 2, pad 1 — native CDNA4 kernels on
    GPU (the last MIOpen ops on the NASNet cell hot path, K9 remainder);
    torch pooling on CPU."""

    def __init__(self, kind: str, stride: int):
        super().__init__()
        assert kind in ("avg", "max")
        self.kind = kind
        self.stride = stride

    def forward(self, x):
        if x.is_cuda:
            return _Pool3Fn.apply(x.to(torch.bfloat16), self.stride,
                                  self.kind == "max")
        if self.kind == "avg":
            return F.avg_pool2d(x, 3, self.stride, 1,
                                count_include_pad=False)
        return F.max_pool2d(x, 3, self.stride, 1)

    def extra_repr(self):
        return "%s3x3 s%d" % (self.kind, self.stride)
