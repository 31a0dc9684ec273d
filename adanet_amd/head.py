"""Heads: loss / predictions / metrics per task type.

The reference delegates to tf.estimator heads (created in user code and
consumed at adanet/core/ensemble_builder.py:571-583). adanet_amd ships its
own minimal heads built on the fused kernels: MultiClassHead runs the K2
fused softmax-xent; metrics run on-device (K10 argmax_correct) and stream
into python accumulators.
"""

from __future__ import annotations

import abc
from typing import Dict, Optional

import torch
import torch.nn.functional as F

from adanet_amd.ops import _extension
from adanet_amd.ops.xent import softmax_xent


class Head(abc.ABC):

    @property
    @abc.abstractmethod
    def logits_dimension(self) -> int:
        ...

    @abc.abstractmethod
    def loss(self, logits: torch.Tensor, labels: torch.Tensor) -> torch.Tensor:
        """Scalar fp32 training loss."""

    @abc.abstractmethod
    def predictions(self, logits: torch.Tensor) -> Dict[str, torch.Tensor]:
        ...

    def metrics(self, logits: torch.Tensor,
                labels: torch.Tensor) -> Dict[str, float]:
        with torch.no_grad():
            return {
                "average_loss": float(self.loss(logits, labels).detach().cpu())
            }


class MultiClassHead(Head):
    """Softmax cross-entropy head with optional label smoothing
    (reference usage: research/improve_nas/trainer/cifar10.py:149-152
    multi-class head with SUM_OVER_BATCH_SIZE reduction = mean)."""

    def __init__(self, n_classes: int, label_smoothing: float = 0.0):
        if n_classes < 2:
            raise ValueError("n_classes must be >= 2")
        self._n_classes = n_classes
        self._label_smoothing = label_smoothing

    @property
    def logits_dimension(self) -> int:
        return self._n_classes

    def loss(self, logits, labels):
        return softmax_xent(logits, labels.long(),
                            label_smoothing=self._label_smoothing,
                            reduction="mean")

    def predictions(self, logits):
        with torch.no_grad():
            probs = torch.softmax(logits.float(), dim=-1)
            return {
                "logits": logits,
                "probabilities": probs,
                "class_ids": probs.argmax(dim=-1),
            }

    def metrics(self, logits, labels):
        with torch.no_grad():
            labels = labels.long()
            if logits.is_cuda:
                ext = _extension.require()
                correct = torch.zeros((1,), device=logits.device,
                                      dtype=torch.int32)
                ext.argmax_correct(logits.to(torch.bfloat16), labels, None,
                                   correct)
                acc = float(correct.item()) / logits.shape[0]
            else:
                acc = float((logits.argmax(dim=-1) == labels).float().mean())
            return {
                "accuracy": acc,
                "average_loss": float(self.loss(logits, labels).detach().cpu()),
            }


class BinaryClassHead(Head):
    """Single-logit sigmoid head."""

    def __init__(self):
        pass

    @property
    def logits_dimension(self) -> int:
        return 1

    def loss(self, logits, labels):
        return F.binary_cross_entropy_with_logits(
            logits.float().reshape(-1), labels.float().reshape(-1))

    def predictions(self, logits):
        with torch.no_grad():
            p = torch.sigmoid(logits.float()).reshape(-1)
            return {"logits": logits, "probabilities": p,
                    "class_ids": (p > 0.5).long()}

    def metrics(self, logits, labels):
        from adanet_amd.core.eval_metrics import AUCAccumulator
        with torch.no_grad():
            p = torch.sigmoid(logits.float()).reshape(-1)
            y = labels.long().reshape(-1)
            acc = float(((p > 0.5).long() == y).float().mean())
            auc = AUCAccumulator()
            auc.update(p, y)
            out = {"accuracy": acc,
                   "average_loss": float(self.loss(logits, labels).cpu())}
            out.update(auc.value())
            return out


class MultiHead(Head):
    """Composite head over named sub-heads (the reference's multi-head
    support, adanet/core/estimator_test.py:1517: logits split across heads,
    labels a dict keyed by head name, losses summed)."""

    def __init__(self, heads: dict):
        if not heads:
            raise ValueError("heads must not be empty")
        self._heads = dict(heads)
        self._order = sorted(heads.keys())

    @property
    def logits_dimension(self) -> int:
        return sum(self._heads[k].logits_dimension for k in self._order)

    def _split(self, logits):
        out = {}
        off = 0
        for k in self._order:
            d = self._heads[k].logits_dimension
            out[k] = logits[:, off:off + d]
            off += d
        return out

    def loss(self, logits, labels):
        parts = self._split(logits)
        total = None
        for k in self._order:
            term = self._heads[k].loss(parts[k], labels[k])
            total = term if total is None else total + term
        return total

    def predictions(self, logits):
        parts = self._split(logits)
        out = {}
        for k in self._order:
            for name, v in self._heads[k].predictions(parts[k]).items():
                out["%s/%s" % (k, name)] = v
        return out

    def metrics(self, logits, labels):
        parts = self._split(logits)
        out = {}
        for k in self._order:
            for name, v in self._heads[k].metrics(parts[k],
                                                  labels[k]).items():
                out["%s/%s" % (k, name)] = v
        out["average_loss"] = float(self.loss(logits, labels).detach().cpu())
        return out


class RegressionHead(Head):
    """Mean-squared-error head (the reference tests' regression head,
    adanet/core/testing_utils.py:236)."""

    def __init__(self, label_dimension: int = 1):
        self._dim = label_dimension

    @property
    def logits_dimension(self) -> int:
        return self._dim

    def loss(self, logits, labels):
        return F.mse_loss(logits.float().reshape(labels.shape),
                          labels.float())

    def predictions(self, logits):
        return {"predictions": logits}

    def metrics(self, logits, labels):
        with torch.no_grad():
            return {"average_loss": float(self.loss(logits, labels).cpu())}
