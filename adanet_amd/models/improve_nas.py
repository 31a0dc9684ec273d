"""improve_nas: the NASNet CIFAR AdaNet search space.

Re-implements the reference research workload
(research/improve_nas/trainer/improve_nas.py + adanet_improve_nas.py) on
the MI355X-native stack:

  * Builder wraps NasNetCIFAR with knowledge distillation added to the
    subnetwork loss — ADAPTIVE distills the previous ensemble, BORN_AGAIN
    the previous subnetwork (reference improve_nas.py:41-59, 166-181).
  * Generator ensembles a fixed architecture (reference :217);
    DynamicGenerator grows each iteration: one candidate 3 cells deeper
    and one 10 conv-filters wider (reference :266-338).
  * Optimizers: fused momentum-SGD with cosine LR (reference
    optimizer.py:45-135; default hparams adanet_improve_nas.py:181-216:
    num_cells=3, num_conv_filters=10, momentum + cosine lr=0.025,
    weight_decay=5e-4, label_smoothing=0.1, knowledge_distillation=adaptive,
    boosting_iterations=3, force_grow=True, scalar mixture weights).
"""

from __future__ import annotations

import dataclasses
import functools
import math
from typing import List, Optional

import torch
import torch.nn.functional as F

from adanet_amd.models.nasnet import NasNetCIFAR
from adanet_amd.ops.optim import CosineLR, FusedSGD
from adanet_amd.subnetwork.generator import Builder, Generator, Subnetwork
from adanet_amd.subnetwork.report import Report


class KnowledgeDistillation(object):
    """Reference improve_nas.py:41-59."""

    NONE = "none"
    ADAPTIVE = "adaptive"
    BORN_AGAIN = "born_again"


@dataclasses.dataclass
class Hparams:
    """Defaults per reference adanet_improve_nas.py:181-216."""

    num_cells: int = 3
    num_conv_filters: int = 10
    learning_rate: float = 0.025
    momentum: float = 0.9
    weight_decay: float = 5e-4
    label_smoothing: float = 0.1
    knowledge_distillation: str = KnowledgeDistillation.ADAPTIVE
    boosting_iterations: int = 3
    force_grow: bool = True
    learn_mixture_weights: bool = False
    drop_path_keep: float = 0.9
    train_steps: int = 10000  # cosine horizon per iteration
    #: round filter counts up to a multiple of this so the native conv
    #: stack (batched-MFMA pointwise / unfold-GEMM adaptors, ops/conv.py)
    #: engages; 1 = exact reference filter counts (library conv fallback
    #: for unaligned channels). 32 is the GEMM alignment contract.
    channel_multiple: int = 1


class NasNetBuilder(Builder):
    """One NASNet-A CIFAR candidate (reference improve_nas.py:62-215)."""

    def __init__(self, hparams: Hparams, num_cells: Optional[int] = None,
                 num_conv_filters: Optional[int] = None,
                 name_suffix: str = "", seed: Optional[int] = None):
        self._hp = hparams
        self._num_cells = num_cells or hparams.num_cells
        f = num_conv_filters or hparams.num_conv_filters
        m = max(1, int(getattr(hparams, "channel_multiple", 1)))
        self._filters = (f + m - 1) // m * m
        self._suffix = name_suffix
        self._seed = seed

    @property
    def name(self) -> str:
        return "nasnet_a_{}x{}{}".format(self._num_cells, self._filters,
                                         self._suffix)

    def build_subnetwork(self, features, logits_dimension, training,
                         previous_ensemble=None) -> Subnetwork:
        if self._seed is not None:
            torch.manual_seed(self._seed + self._num_cells * 131 +
                              self._filters)
        module = NasNetCIFAR(
            num_cells=self._num_cells,
            num_conv_filters=self._filters,
            num_classes=logits_dimension,
            drop_path_keep=self._hp.drop_path_keep if training else 1.0)
        # complexity ~ sqrt(parameter count in millions): deeper/wider
        # candidates pay a larger mixture-weight penalty.
        n_params = sum(p.numel() for p in module.parameters())
        return Subnetwork(module=module,
                          complexity=math.sqrt(n_params / 1e6),
                          shared={"num_cells": self._num_cells,
                                  "num_conv_filters": self._filters},
                          name=self.name)

    def build_optimizer(self, params, iteration: int = 0):
        opt = FusedSGD(params, lr=self._hp.learning_rate,
                       momentum=self._hp.momentum,
                       weight_decay=self._hp.weight_decay)
        opt._adanet_lr_sched = CosineLR(opt, self._hp.train_steps)
        return opt

    def build_subnetwork_loss(self, head, logits, labels, features,
                              previous_ensemble_logits_fn,
                              frozen_outputs_fn):
        """Cross-entropy (+label smoothing) + knowledge distillation
        (reference improve_nas.py:160-181: soft cross-entropy against the
        teacher's probabilities added to the base loss)."""
        base = head.loss(logits, labels)
        kd = self._hp.knowledge_distillation
        if kd == KnowledgeDistillation.NONE:
            return base
        teacher = None
        if kd == KnowledgeDistillation.ADAPTIVE:
            teacher = previous_ensemble_logits_fn()
        elif kd == KnowledgeDistillation.BORN_AGAIN:
            fo = frozen_outputs_fn()
            if fo:
                # the most recent frozen subnetwork's logits
                teacher = list(fo.values())[-1][1]
        if teacher is None:
            return base
        with torch.no_grad():
            p_t = torch.softmax(teacher.float(), dim=-1)
        log_q = torch.log_softmax(logits.float(), dim=-1)
        distill = -(p_t * log_q).sum(dim=-1).mean()
        return base + distill

    def build_subnetwork_report(self) -> Report:
        return Report(hparams={"num_cells": self._num_cells,
                               "num_conv_filters": self._filters},
                      attributes={}, metrics={})


class Generator(Generator):
    """Fixed-architecture ensembling (reference improve_nas.py:217-264)."""

    def __init__(self, hparams: Optional[Hparams] = None,
                 seed: Optional[int] = None):
        self._hp = hparams or Hparams()
        self._seed = seed

    def generate_candidates(self, previous_ensemble, iteration_number,
                            previous_ensemble_reports, all_reports,
                            config=None) -> List[Builder]:
        return [NasNetBuilder(self._hp, seed=self._seed)]


class DynamicGenerator(Generator):
    """Each iteration: one candidate 3 cells deeper and one 10 filters
    wider than the previous best (reference improve_nas.py:266-338)."""

    def __init__(self, hparams: Optional[Hparams] = None,
                 seed: Optional[int] = None):
        super().__init__(hparams, seed)

    def generate_candidates(self, previous_ensemble, iteration_number,
                            previous_ensemble_reports, all_reports,
                            config=None) -> List[Builder]:
        cells = self._hp.num_cells
        filters = self._hp.num_conv_filters
        if previous_ensemble is not None:
            shared = previous_ensemble.subnetworks[-1].shared or {}
            cells = int(shared.get("num_cells", cells))
            filters = int(shared.get("num_conv_filters", filters))
        return [
            NasNetBuilder(self._hp, num_cells=cells + 3,
                          num_conv_filters=filters, name_suffix="_deeper",
                          seed=self._seed),
            NasNetBuilder(self._hp, num_cells=cells,
                          num_conv_filters=filters + 10,
                          name_suffix="_wider", seed=self._seed),
        ]
