"""Fused weighted-sum-of-logits ensemble mixer (K5).

out[b,c] = bias[c] + sum_j w_j (*) L_j[b,c] over the J member logit buffers
in ONE kernel (csrc/mixer.hip); the frozen members' logits come from the
iteration's HBM cache (no recompute). Autograd flows to the weights, the
bias, and to any member logits that require grad (the newly-training
subnetwork); frozen members get no gradient buffers at all.

Reference behavior: adanet/ensemble/weighted.py:427-454,545-561.
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import torch

from adanet_amd.ops import _extension


class _MixerFn(torch.autograd.Function):

    @staticmethod
    def forward(ctx, stacked_w, bias, vector_mode, *logits):
        ext = _extension.require()
        B, C = logits[0].shape
        # Direct-pointer path: the kernel reads the J member buffers in
        # place (frozen members' static HBM-cache buffers + the live
        # candidate's logits) — no per-step [J,B,C] stack copy. Pointers
        # live in the kernel args, so hipGraph capture bakes the stable
        # static-buffer addresses (unlike the old host-built pointer
        # table whose H2D copy captured stale addresses).
        members = [l.detach() for l in logits]
        out = torch.empty((B, C), device=logits[0].device,
                          dtype=torch.bfloat16)
        ext.mixer_fwd_direct(members, stacked_w, bias, out,
                             1 if vector_mode else 0)
        ctx.vector_mode = vector_mode
        ctx.has_bias = bias is not None
        ctx.logit_requires = [t.requires_grad for t in logits]
        # live Parameter identities for direct-to-arena grad writes
        # (mixture phase runs under ops.linear.direct_grad_writes).
        ctx.param_refs = (
            stacked_w if isinstance(stacked_w, torch.nn.Parameter) else None,
            bias if isinstance(bias, torch.nn.Parameter) else None)
        ctx.save_for_backward(stacked_w, *members)
        return out

    @staticmethod
    def backward(ctx, dy):
        from adanet_amd.ops import linear as _lin
        ext = _extension.require()
        stacked_w = ctx.saved_tensors[0]
        members = list(ctx.saved_tensors[1:])
        B, C = members[0].shape
        dy = dy.contiguous()
        wparam, bparam = ctx.param_refs
        # Direct-to-arena: the mixture weights are used once per step, so
        # the scalar kernel's atomicAdds (into the zeroed arena view) /
        # the vector kernel's overwrites land the final gradient and we
        # return None — no AccumulateGrad add, no zeros_like fill.
        wg = wparam.grad if (_lin._DIRECT_GRAD and wparam is not None) \
            else None
        direct_w = (wg is not None and wg.is_cuda
                    and wg.dtype == torch.float32
                    and wg.shape == stacked_w.shape and wg.is_contiguous())
        dw_out = wg if direct_w else torch.zeros_like(stacked_w)
        ext.mixer_bwd_dw_direct(members, dy, dw_out,
                                1 if ctx.vector_mode else 0)
        dw = None if direct_w else dw_out
        dbias = None
        if ctx.has_bias:
            bg = bparam.grad if (_lin._DIRECT_GRAD and bparam is not None) \
                else None
            direct_b = (bg is not None and bg.is_cuda
                        and bg.dtype == torch.float32
                        and bg.shape == (C,) and bg.is_contiguous())
            if direct_b:
                ext.colsum_bf16(dy, bg, 1)
            else:
                dbias = torch.empty((C,), device=dy.device,
                                    dtype=torch.float32)
                ext.colsum_bf16(dy, dbias)
        dlogits = []
        for j, req in enumerate(ctx.logit_requires):
            if req:
                dl = torch.empty((B, C), device=dy.device,
                                 dtype=torch.bfloat16)
                ext.mixer_bwd_dlogits(dy, stacked_w, dl, j,
                                      1 if ctx.vector_mode else 0)
                dlogits.append(dl)
            else:
                dlogits.append(None)
        return (dw, dbias, None) + tuple(dlogits)


def weighted_sum_logits(logits: Sequence[torch.Tensor],
                        weights,
                        bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """bias + sum_j w_j * logits_j with scalar or per-class vector weights.

    ``weights`` is either the ensemble's FLAT parameter ([J] scalar mode /
    [J, C] vector mode — the zero-copy fast path) or a list of per-member
    weight tensors (stacked here; autograd un-stacks dw back to each).
    """
    assert len(logits) > 0
    if torch.is_tensor(weights):
        stacked = weights
        assert stacked.shape[0] == len(logits)
        vector_mode = stacked.dim() == 2 and stacked.shape[1] > 1
    else:
        assert len(logits) == len(weights)
        vector_mode = weights[0].dim() >= 1 and weights[0].numel() > 1
        stacked = torch.stack([w.reshape(-1) for w in weights])
        if not vector_mode:
            stacked = stacked.reshape(len(weights))
    if logits[0].is_cuda:
        stacked = stacked.contiguous()
        logits = [l if l.dtype == torch.bfloat16 else l.to(torch.bfloat16)
                  for l in logits]
        return _MixerFn.apply(stacked, bias, vector_mode, *logits)
    # CPU reference path (fp32).
    total = None
    for j, l in enumerate(logits):
        w = stacked[j]
        term = l.float() * w.float()
        total = term if total is None else total + term
    if bias is not None:
        total = total + bias.float()
    return total.to(logits[0].dtype)
