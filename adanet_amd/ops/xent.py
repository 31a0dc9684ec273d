"""Fused softmax cross-entropy (K2) with label smoothing.

GPU: one-pass fused kernel producing per-row fp32 loss + bf16 softmax probs
(csrc/softmax_xent.hip); backward is closed-form (p - target) * grad.
CPU: fp32 torch reference (the numerics oracle used by tests/).

Replaces the reference head loss (adanet/core/ensemble_builder.py:571-583,
tf.losses.softmax_cross_entropy with label_smoothing in
research/improve_nas/trainer/improve_nas.py:160-181).
"""

from __future__ import annotations

import torch
import torch.nn.functional as F

from adanet_amd.ops import _extension


class _SoftmaxXentFn(torch.autograd.Function):
    """Per-row loss vector variant (reduction='none'/'sum')."""

    @staticmethod
    def forward(ctx, logits, labels, label_smoothing):
        ext = _extension.require()
        B, C = logits.shape
        loss = torch.empty((B,), device=logits.device, dtype=torch.float32)
        probs = torch.empty((B, C), device=logits.device,
                            dtype=torch.bfloat16)
        ext.softmax_xent_fwd(logits, labels, loss, probs,
                             float(label_smoothing), None)
        ctx.save_for_backward(probs, labels)
        ctx.label_smoothing = float(label_smoothing)
        return loss

    @staticmethod
    def backward(ctx, grad_rows):
        probs, labels = ctx.saved_tensors
        ext = _extension.require()
        dlogits = torch.empty_like(probs)
        ext.softmax_xent_bwd(probs, labels, grad_rows.contiguous().float(),
                             dlogits, ctx.label_smoothing, None)
        return dlogits, None, None


class _SoftmaxXentMeanFn(torch.autograd.Function):
    """Fused-mean variant: forward returns the scalar mean loss directly
    (one kernel, no torch reduce), backward reads the upstream scalar grad
    from device memory (hipGraph-safe, no host scalars)."""

    @staticmethod
    def forward(ctx, logits, labels, label_smoothing):
        ext = _extension.require()
        B, C = logits.shape
        # empty, not zeros: the partials+reducer path overwrites the scalar
        # (no per-step fill kernel in the captured graph).
        mean = torch.empty((), device=logits.device, dtype=torch.float32)
        probs = torch.empty((B, C), device=logits.device,
                            dtype=torch.bfloat16)
        ext.softmax_xent_fwd(logits, labels, None, probs,
                             float(label_smoothing), mean)
        ctx.save_for_backward(probs, labels)
        ctx.label_smoothing = float(label_smoothing)
        return mean

    @staticmethod
    def backward(ctx, grad):
        probs, labels = ctx.saved_tensors
        ext = _extension.require()
        dlogits = torch.empty_like(probs)
        ext.softmax_xent_bwd(probs, labels, None, dlogits,
                             ctx.label_smoothing,
                             grad.contiguous().float())
        return dlogits, None, None


def softmax_xent(logits: torch.Tensor, labels: torch.Tensor,
                 label_smoothing: float = 0.0,
                 reduction: str = "mean") -> torch.Tensor:
    """Per-row (or reduced) softmax cross-entropy loss, fp32."""
    if logits.is_cuda:
        if logits.stride(1) != 1:
            logits = logits.contiguous()
        if reduction == "mean":
            return _SoftmaxXentMeanFn.apply(logits, labels, label_smoothing)
        per_row = _SoftmaxXentFn.apply(logits, labels, label_smoothing)
    else:
        per_row = F.cross_entropy(logits.float(), labels,
                                  label_smoothing=label_smoothing,
                                  reduction="none")
        if reduction == "mean":
            return per_row.mean()
    if reduction == "sum":
        return per_row.sum()
    return per_row
