"""Fused optimizers (K4): SGD(+momentum/nesterov) and Adam.

bf16 parameters keep an fp32 master copy in optimizer state; the GPU update
is ONE kernel pass per tensor that updates the master, refreshes the bf16
working copy, and applies momentum/Adam state (csrc/optim.hip). fp32
parameters (e.g. mixture weights, biases — tiny) update in eager torch math
on either device.

These are drop-in ``torch.optim.Optimizer`` subclasses so user Builders can
return them from ``build_optimizer`` (the analog of the reference's
``build_subnetwork_train_op``, adanet/subnetwork/generator.py:198-229; the
optimizer registry it replaces: research/improve_nas/trainer/optimizer.py:
104-131).
"""

from __future__ import annotations

import math
from typing import Iterable, Optional

import torch

from adanet_amd.ops import _extension


def _fp32_sgd_update(p, g, state, lr, momentum, dampening, weight_decay,
                     nesterov):
    master = state.get("master")
    if master is None:
        master = p.data.float().clone()
        state["master"] = master
    gf = g.float()
    if weight_decay:
        gf = gf + weight_decay * master
    if momentum:
        buf = state.get("momentum_buffer")
        if buf is None:
            buf = torch.zeros_like(master)
            state["momentum_buffer"] = buf
        buf.mul_(momentum).add_(gf, alpha=1.0 - dampening)
        gf = gf + momentum * buf if nesterov else buf
    master.add_(gf, alpha=-lr)
    p.data.copy_(master.to(p.dtype))


class FusedSGD(torch.optim.Optimizer):
    """SGD with momentum/nesterov; fused single-pass kernel for bf16 params."""

    def __init__(self, params, lr: float = 0.01, momentum: float = 0.0,
                 dampening: float = 0.0, weight_decay: float = 0.0,
                 nesterov: bool = False):
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError("Nesterov momentum requires momentum>0, dampening=0")
        defaults = dict(lr=lr, momentum=momentum, dampening=dampening,
                        weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if p.is_cuda and p.dtype == torch.bfloat16:
                    ext = _extension.require()
                    master = state.get("master")
                    if master is None:
                        master = p.data.float().contiguous()
                        state["master"] = master
                    mom = None
                    if group["momentum"]:
                        mom = state.get("momentum_buffer")
                        if mom is None:
                            mom = torch.zeros_like(master)
                            state["momentum_buffer"] = mom
                    grad = p.grad
                    if grad.dtype != torch.bfloat16:
                        grad = grad.to(torch.bfloat16)
                    ext.fused_sgd(master, p.data, grad.contiguous(), mom,
                                  group["lr"], group["momentum"],
                                  group["dampening"], group["weight_decay"],
                                  group["nesterov"], 1.0)
                elif p.is_cuda and p.dtype == torch.float32:
                    ext = _extension.require()
                    mom = None
                    if group["momentum"]:
                        mom = state.get("momentum_buffer")
                        if mom is None:
                            mom = torch.zeros_like(p.data)
                            state["momentum_buffer"] = mom
                    ext.fused_sgd_fp32(p.data,
                                       p.grad.float().contiguous(), mom,
                                       group["lr"], group["momentum"],
                                       group["dampening"],
                                       group["weight_decay"],
                                       group["nesterov"], 1.0)
                else:
                    _fp32_sgd_update(p, p.grad, state, group["lr"],
                                     group["momentum"], group["dampening"],
                                     group["weight_decay"], group["nesterov"])
        return loss


class FusedAdam(torch.optim.Optimizer):
    """Adam; fused single-pass kernel for bf16 params (fp32 master + m/v)."""

    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            b1, b2 = group["betas"]
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                state["step"] = state.get("step", 0) + 1
                t = state["step"]
                if p.is_cuda and p.dtype == torch.bfloat16:
                    ext = _extension.require()
                    if "master" not in state:
                        state["master"] = p.data.float().contiguous()
                        state["exp_avg"] = torch.zeros_like(state["master"])
                        state["exp_avg_sq"] = torch.zeros_like(state["master"])
                    grad = p.grad
                    if grad.dtype != torch.bfloat16:
                        grad = grad.to(torch.bfloat16)
                    ext.fused_adam(state["master"], p.data, grad.contiguous(),
                                   state["exp_avg"], state["exp_avg_sq"],
                                   group["lr"], b1, b2, group["eps"],
                                   group["weight_decay"], t, 1.0)
                elif p.is_cuda and p.dtype == torch.float32:
                    ext = _extension.require()
                    if "exp_avg" not in state:
                        state["exp_avg"] = torch.zeros_like(p.data)
                        state["exp_avg_sq"] = torch.zeros_like(p.data)
                    ext.fused_adam_fp32(p.data, p.grad.float().contiguous(),
                                        state["exp_avg"],
                                        state["exp_avg_sq"], group["lr"], b1,
                                        b2, group["eps"],
                                        group["weight_decay"], t, 1.0)
                else:
                    master = state.get("master")
                    if master is None:
                        master = p.data.float().clone()
                        state["master"] = master
                        state["exp_avg"] = torch.zeros_like(master)
                        state["exp_avg_sq"] = torch.zeros_like(master)
                    g = p.grad.float()
                    if group["weight_decay"]:
                        g = g + group["weight_decay"] * master
                    m, v = state["exp_avg"], state["exp_avg_sq"]
                    m.mul_(b1).add_(g, alpha=1 - b1)
                    v.mul_(b2).addcmul_(g, g, value=1 - b2)
                    mhat = m / (1 - b1 ** t)
                    vhat = v / (1 - b2 ** t)
                    master.addcdiv_(mhat, vhat.sqrt().add_(group["eps"]),
                                    value=-group["lr"])
                    p.data.copy_(master.to(p.dtype))
        return loss


def make_optimizer(name: str, params, lr: float, **kwargs):
    """Optimizer registry (reference research/improve_nas/trainer/
    optimizer.py:104-131: adagrad/adam/lazy_adam/momentum/rmsprop/sgd)."""
    name = name.lower()
    if name == "sgd":
        return FusedSGD(params, lr=lr, **kwargs)
    if name == "momentum":
        kwargs.setdefault("momentum", 0.9)
        return FusedSGD(params, lr=lr, **kwargs)
    if name in ("adam", "lazy_adam"):
        return FusedAdam(params, lr=lr, **kwargs)
    if name == "adagrad":
        return torch.optim.Adagrad(params, lr=lr, **kwargs)
    if name == "rmsprop":
        return torch.optim.RMSprop(params, lr=lr, **kwargs)
    raise ValueError("unknown optimizer %r" % (name,))


class CosineLR(object):
    """Cosine decay schedule (reference trainer/optimizer.py:45-103
    cosine_decay used with momentum for NASNet)."""

    def __init__(self, optimizer, total_steps: int, min_factor: float = 0.0):
        self.optimizer = optimizer
        self.total_steps = max(1, total_steps)
        self.min_factor = min_factor
        self.base_lrs = [g["lr"] for g in optimizer.param_groups]
        self.step_num = 0

    def step(self):
        self.step_num += 1
        t = min(self.step_num / self.total_steps, 1.0)
        factor = self.min_factor + (1 - self.min_factor) * 0.5 * (
            1 + math.cos(math.pi * t))
        for g, base in zip(self.optimizer.param_groups, self.base_lrs):
            g["lr"] = base * factor
