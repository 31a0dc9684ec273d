"""CompiledModel: a keras-like fit/evaluate wrapper over nn.Module.

The reference's ModelFlow drives keras models through model.fit/evaluate
(adanet/experimental/work_units/keras_trainer_work_unit.py:41-55). The
MI355X-native equivalent wraps an nn.Module + optimizer + loss into the
same two-method surface so phases/work-units stay structurally identical.
"""

from __future__ import annotations

from typing import Callable, Dict, Iterable, Optional

import torch
from torch import nn


class CompiledModel(object):

    def __init__(self, module: nn.Module, optimizer=None,
                 loss_fn: Optional[Callable] = None,
                 metric_fn: Optional[Callable] = None,
                 device: Optional[torch.device] = None):
        self.module = module
        self.device = device or (torch.device("cuda:0")
                                 if torch.cuda.is_available()
                                 else torch.device("cpu"))
        self.module.to(self.device)
        if self.device.type == "cuda":
            self.module.to(torch.bfloat16)
            from adanet_amd.ops.linear import restore_fp32_params
            restore_fp32_params(self.module)
        if optimizer is None:
            from adanet_amd.ops.optim import FusedSGD
            params = list(self.module.parameters())
            optimizer = FusedSGD(params, lr=0.01) if params else None
        elif callable(optimizer) and not hasattr(optimizer, "step"):
            optimizer = optimizer(self.module.parameters())
        self.optimizer = optimizer
        self.loss_fn = loss_fn or self._default_loss
        self.metric_fn = metric_fn

    @staticmethod
    def _default_loss(logits, labels):
        from adanet_amd.ops.xent import softmax_xent
        if logits.shape[-1] == 1 or logits.dim() == 1:
            return torch.nn.functional.mse_loss(
                logits.float().reshape(labels.shape), labels.float())
        return softmax_xent(logits, labels.long())

    def _to_dev(self, x, y):
        x = x.to(self.device)
        if x.is_floating_point() and self.device.type == "cuda":
            x = x.to(torch.bfloat16)
        return x, (y.to(self.device) if y is not None else None)

    def __call__(self, x):
        out = self.module(x)
        return out[1] if isinstance(out, tuple) else out

    def fit(self, dataset: Iterable, epochs: int = 1,
            steps_per_epoch: Optional[int] = None) -> Dict[str, float]:
        self.module.train()
        last = float("nan")
        for _ in range(epochs):
            for i, (x, y) in enumerate(dataset):
                if steps_per_epoch is not None and i >= steps_per_epoch:
                    break
                x, y = self._to_dev(x, y)
                logits = self(x)
                loss = self.loss_fn(logits, y)
                if self.optimizer is not None:
                    self.optimizer.zero_grad(set_to_none=True)
                    loss.backward()
                    self.optimizer.step()
                last = float(loss.detach())
        return {"loss": last}

    def evaluate(self, dataset: Iterable,
                 steps: Optional[int] = None) -> Dict[str, float]:
        self.module.eval()
        total, n = 0.0, 0
        correct = 0
        examples = 0
        with torch.no_grad():
            for i, (x, y) in enumerate(dataset):
                if steps is not None and i >= steps:
                    break
                x, y = self._to_dev(x, y)
                logits = self(x)
                total += float(self.loss_fn(logits, y))
                n += 1
                if logits.dim() == 2 and logits.shape[-1] > 1:
                    correct += int((logits.float().argmax(-1) ==
                                    y.long()).sum())
                    examples += int(y.numel())
        out = {"loss": total / max(n, 1)}
        if examples:
            out["accuracy"] = correct / examples
        return out
