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


class _ArenaMixin(object):
    """Transparent parameter-arena flattening for the fused optimizers.

    On the first GPU step, each param group's CUDA parameters coalesce by
    dtype into ONE contiguous working buffer (plus flat master / momentum /
    Adam state); `p.data` and `p.grad` become views, so:

      * the whole update is one fused kernel per (group, dtype) instead of
        one per tensor,
      * data-parallel gradient all-reduce is ONE zero-copy xGMI bucket per
        candidate (`flat_grad_buffers()` — used by distributed/comm.py),
      * `zero_grad` is one memset per buffer (and re-pins the grad views,
        so the engine's `set_to_none=True` is safely ignored).

    The arena is built lazily after the first backward (grads must exist to
    seed the views), which is also before any hipGraph capture (warmup
    steps), so the captured graph sees only stable flat buffers.
    """

    def _arena_init(self):
        self._arenas = None  # list of dicts per (group, dtype)

    def _maybe_flatten(self):
        if self._arenas is not None:
            return
        arenas = []
        for gi, group in enumerate(self.param_groups):
            by_dtype = {}
            for p in group["params"]:
                if not p.is_cuda or p.grad is None:
                    continue
                by_dtype.setdefault(p.dtype, []).append(p)
            for dtype, params in by_dtype.items():
                if len(params) < 2:
                    continue
                n = sum(p.numel() for p in params)
                device = params[0].device
                flat_p = torch.empty(n, device=device, dtype=dtype)
                flat_g = torch.zeros(n, device=device, dtype=dtype)
                off = 0
                views = []
                for p in params:
                    k = p.numel()
                    flat_p[off:off + k] = p.data.reshape(-1)
                    flat_g[off:off + k] = p.grad.reshape(-1).to(dtype)
                    p.data = flat_p[off:off + k].view_as(p.data)
                    p.grad = flat_g[off:off + k].view_as(p.data)
                    views.append((p, off, k))
                    # migrate any per-param state already created
                    self.state.pop(p, None)
                    off += k
                arenas.append({
                    "group": gi, "dtype": dtype, "params": params,
                    "flat_p": flat_p, "flat_g": flat_g, "views": views,
                    "state": {},
                    # every param single-write (HipLinear weights): with
                    # overwrite-mode direct grad writes the per-step arena
                    # zero fill can be skipped entirely.
                    "all_single_write": all(
                        getattr(p, "_adanet_single_write", False)
                        for p in params),
                })
        self._arenas = arenas

    def _arena_params(self):
        covered = set()
        for a in self._arenas or []:
            for p in a["params"]:
                covered.add(id(p))
        return covered

    def flat_grad_buffers(self):
        """Flat gradient buckets for zero-copy all-reduce (+ leftover
        per-param grads)."""
        bufs = []
        if self._arenas:
            for a in self._arenas:
                bufs.append(a["flat_g"])
        covered = self._arena_params()
        for group in self.param_groups:
            for p in group["params"]:
                if id(p) not in covered and p.grad is not None:
                    bufs.append(p.grad)
        return bufs

    def zero_grad(self, set_to_none: bool = True):
        if self._arenas:
            for a in self._arenas:
                # Overwrite-mode (engine opt-in via _overwrite_grads, only
                # when every param in the arena is written exactly once
                # per backward by a direct-grad producer): skip the memset
                # and tell producers to use the overwrite epilogue.
                skip = (getattr(self, "_overwrite_grads", False)
                        and a["all_single_write"])
                if not skip:
                    a["flat_g"].zero_()
                for p, off, k in a["views"]:
                    p._adanet_grad_overwrite = skip
                    p._adanet_bwd_writes = 0
                    if p.grad is None or p.grad.data_ptr() != a[
                            "flat_g"][off:off + k].data_ptr():
                        p.grad = a["flat_g"][off:off + k].view_as(p.data)
            covered = self._arena_params()
            for group in self.param_groups:
                for p in group["params"]:
                    if id(p) not in covered and p.grad is not None:
                        if set_to_none:
                            p.grad = None
                        else:
                            p.grad.zero_()
            return
        super().zero_grad(set_to_none=set_to_none)

    def _arena_state_dict(self):
        out = []
        for a in self._arenas or []:
            out.append({
                "group": a["group"], "dtype": str(a["dtype"]),
                "flat_p": a["flat_p"].detach().cpu(),
                "state": {k: (v.detach().cpu() if torch.is_tensor(v) else v)
                          for k, v in a["state"].items()},
            })
        return out

    def _arena_load_state_dict(self, blobs):
        if not blobs or self._arenas is None:
            return False
        if len(blobs) != len(self._arenas):
            return False
        for a, blob in zip(self._arenas, blobs):
            a["flat_p"].copy_(blob["flat_p"].to(a["flat_p"].device))
            for k, v in blob["state"].items():
                a["state"][k] = (v.to(a["flat_p"].device)
                                 if torch.is_tensor(v) else v)
        return True

    def state_dict(self):
        sd = super().state_dict()
        if self._arenas:
            sd["_arenas"] = self._arena_state_dict()
        return sd

    def load_state_dict(self, sd):
        blobs = sd.pop("_arenas", None)
        try:
            super().load_state_dict(sd)
        except Exception:
            pass  # per-param state may not round-trip across arena builds
        if blobs is not None:
            if self._arenas is None:
                self._pending_arena_state = blobs
            else:
                self._arena_load_state_dict(blobs)


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


class FusedSGD(_ArenaMixin, torch.optim.Optimizer):
    """SGD with momentum/nesterov; fused single-pass kernel for bf16 params
    with automatic parameter-arena flattening (see _ArenaMixin)."""

    def __init__(self, params, lr: float = 0.01, momentum: float = 0.0,
                 dampening: float = 0.0, weight_decay: float = 0.0,
                 nesterov: bool = False):
        if nesterov and (momentum <= 0 or dampening != 0):
            raise ValueError("Nesterov momentum requires momentum>0, dampening=0")
        defaults = dict(lr=lr, momentum=momentum, dampening=dampening,
                        weight_decay=weight_decay, nesterov=nesterov)
        super().__init__(params, defaults)
        self._arena_init()
        self._pending_arena_state = None

    def _arena_step(self, a):
        ext = _extension.require()
        group = self.param_groups[a["group"]]
        state = a["state"]
        mom = None
        if group["momentum"]:
            mom = state.get("momentum_buffer")
            if mom is None:
                mom = torch.zeros(a["flat_p"].numel(),
                                  device=a["flat_p"].device,
                                  dtype=torch.float32)
                state["momentum_buffer"] = mom
        if a["dtype"] == torch.bfloat16:
            master = state.get("master")
            if master is None:
                master = a["flat_p"].float()
                state["master"] = master
            ext.fused_sgd(master, a["flat_p"], a["flat_g"], mom,
                          group["lr"], group["momentum"],
                          group["dampening"], group["weight_decay"],
                          group["nesterov"], 1.0)
        else:
            ext.fused_sgd_fp32(a["flat_p"], a["flat_g"], mom, group["lr"],
                               group["momentum"], group["dampening"],
                               group["weight_decay"], group["nesterov"], 1.0)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        self._maybe_flatten()
        if self._pending_arena_state is not None and self._arenas:
            self._arena_load_state_dict(self._pending_arena_state)
            self._pending_arena_state = None
        covered = self._arena_params()
        for a in self._arenas or []:
            self._arena_step(a)
        for group in self.param_groups:
            for p in group["params"]:
                if p.grad is None or id(p) in covered:
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


class FusedAdam(_ArenaMixin, torch.optim.Optimizer):
    """Adam; fused single-pass kernel for bf16 params (fp32 master + m/v)
    with automatic parameter-arena flattening (see _ArenaMixin)."""

    def __init__(self, params, lr: float = 1e-3, betas=(0.9, 0.999),
                 eps: float = 1e-8, weight_decay: float = 0.0):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self._arena_init()
        self._pending_arena_state = None

    def _arena_step(self, a):
        ext = _extension.require()
        group = self.param_groups[a["group"]]
        b1, b2 = group["betas"]
        state = a["state"]
        state["step"] = state.get("step", 0) + 1
        if "exp_avg" not in state:
            n = a["flat_p"].numel()
            dev = a["flat_p"].device
            state["exp_avg"] = torch.zeros(n, device=dev,
                                           dtype=torch.float32)
            state["exp_avg_sq"] = torch.zeros(n, device=dev,
                                              dtype=torch.float32)
        if a["dtype"] == torch.bfloat16:
            master = state.get("master")
            if master is None:
                master = a["flat_p"].float()
                state["master"] = master
            ext.fused_adam(master, a["flat_p"], a["flat_g"],
                           state["exp_avg"], state["exp_avg_sq"],
                           group["lr"], b1, b2, group["eps"],
                           group["weight_decay"], state["step"], 1.0)
        else:
            ext.fused_adam_fp32(a["flat_p"], a["flat_g"], state["exp_avg"],
                                state["exp_avg_sq"], group["lr"], b1, b2,
                                group["eps"], group["weight_decay"],
                                state["step"], 1.0)

    @torch.no_grad()
    def step(self, closure=None):
        loss = closure() if closure is not None else None
        self._maybe_flatten()
        if self._pending_arena_state is not None and self._arenas:
            self._arena_load_state_dict(self._pending_arena_state)
            self._pending_arena_state = None
        covered = self._arena_params()
        for a in self._arenas or []:
            self._arena_step(a)
        for group in self.param_groups:
            b1, b2 = group["betas"]
            for p in group["params"]:
                if p.grad is None or id(p) in covered:
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
