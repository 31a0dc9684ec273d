"""_Iteration: one AdaNet round — all candidate subnetworks + ensembles
training simultaneously.

Reference: adanet/core/iteration.py (the _IterationBuilder/_Iteration/
_TrainManager machinery, :40-1230) and adanet/core/ensemble_builder.py
(_SubnetworkManager/_EnsembleBuilder, :258-805). The reference builds ONE
TF graph per iteration and gates each candidate's train op with hooks; the
MI355X-native engine is define-by-run:

  * every training step runs each ACTIVE subnetwork's fwd+bwd+fused-optimizer
    on its own HIP stream (overlapping candidates on the 256-CU chip),
  * frozen previous-iteration subnetworks run ONCE per batch under no_grad
    in eval mode (reference "dropout off for frozen members",
    iteration.py:567-579) and their logits are cached in HBM for every
    candidate ensemble's fused mixer call,
  * candidate ensembles train only their mixture weights on the AdaNet
    objective (reference weighted.py:606-617 var_list=mixture weights),
  * per-candidate AdaNet losses accumulate in a device ring buffer and are
    flushed to host EMAs in chunks, so the steady-state step issues no
    host<->device synchronization.
"""

from __future__ import annotations

import dataclasses
import json
import logging
import math
import os
from typing import Callable, Dict, List, Optional, Sequence, Tuple

import torch
from torch import nn

from adanet_amd.core.architecture import _Architecture
from adanet_amd.core.candidate import _Candidate
from adanet_amd.core.summary import _ScopedSummary
from adanet_amd.distributed import comm
from adanet_amd.ensemble.strategy import Candidate as EnsembleCandidate
from adanet_amd.ops.linear import direct_grad_writes
from adanet_amd.subnetwork.generator import Builder, Subnetwork

log = logging.getLogger("adanet_amd")

_LOSS_FLUSH_STEPS = 16


class _FrozenLogitCache(object):
    """HBM-resident cache of frozen-member outputs, keyed
    (batch_cache_key, member_name) -> (last_layer | None, logits).

    Frozen member weights are immutable once frozen, so entries stay valid
    ACROSS AdaNet iterations — only the newly frozen winner of iteration
    t-1 is a miss at iteration t. This is what flattens the per-member
    cost curve: without it every iteration recomputes all J members on
    every resident batch (O(J) forwards per iteration, the dominant growth
    term the round-1 driver measured at large ensembles). Owned by the
    Estimator; _Iteration holds a reference. Insertion stops at `cap`
    bytes (ADANET_FROZEN_CACHE_BYTES, default 32 GiB of the 288 GB HBM).
    """

    def __init__(self, cap_bytes: Optional[int] = None):
        if cap_bytes is None:
            cap_bytes = int(
                os.environ.get("ADANET_FROZEN_CACHE_BYTES",
                               32 * 1024 ** 3))
        self.cap = cap_bytes
        self.bytes = 0
        self._d: Dict = {}

    def get(self, key):
        return self._d.get(key)

    def put(self, key, last, logits):
        sz = logits.numel() * logits.element_size() + (
            last.numel() * last.element_size() if last is not None else 0)
        old = self._d.get(key)
        if old is not None:
            self.bytes -= (old[1].numel() * old[1].element_size() +
                           (old[0].numel() * old[0].element_size()
                            if old[0] is not None else 0))
        if self.bytes + sz > self.cap:
            return
        self._d[key] = (last, logits)
        self.bytes += sz

    def __len__(self):
        return len(self._d)


class _TrainManager(object):
    """Persists per-spec "done training" state as JSON files.

    Reference: adanet/core/iteration.py:40-118 — same on-disk layout
    (<model_dir>/train_manager/t<t>/<spec_name>.json) so restarts and
    preemptions resume correctly; only the chief writes.
    """

    def __init__(self, model_dir: Optional[str], iteration_number: int,
                 is_chief: bool = True):
        self._is_chief = is_chief
        self._dir = None
        self._stopped: Dict[str, str] = {}
        if model_dir:
            self._dir = os.path.join(model_dir, "train_manager",
                                     "t{}".format(iteration_number))
            if is_chief:
                os.makedirs(self._dir, exist_ok=True)
            if os.path.isdir(self._dir):
                for fname in os.listdir(self._dir):
                    if fname.endswith(".json"):
                        name = fname[:-len(".json")]
                        try:
                            with open(os.path.join(self._dir, fname)) as f:
                                payload = json.load(f)
                        except (json.JSONDecodeError, OSError):
                            payload = {}
                        self._stopped[name] = payload.get("reason", "")

    def should_train(self, spec_name: str) -> bool:
        return spec_name not in self._stopped

    def request_stop(self, spec_name: str, reason: str = ""):
        if spec_name in self._stopped:
            return
        self._stopped[spec_name] = reason
        if self._dir and self._is_chief:
            path = os.path.join(self._dir, spec_name + ".json")
            with open(path, "w") as f:
                json.dump({"reason": reason}, f)

    def is_over(self, spec_names: Sequence[str]) -> bool:
        return all(n in self._stopped for n in spec_names)

    @property
    def stopped(self) -> Dict[str, str]:
        return dict(self._stopped)


@dataclasses.dataclass
class _SubnetworkSpec:
    """One candidate subnetwork under training (reference
    adanet/core/ensemble_builder.py:586-641 _SubnetworkSpec)."""

    name: str
    builder: Builder
    subnetwork: Optional[Subnetwork]  # None on non-owner ranks (round-robin)
    optimizer: Optional[object]
    owner_rank: int = 0
    step: int = 0
    last_loss: float = float("nan")
    stream: Optional[object] = None
    summary: Optional[_ScopedSummary] = None
    # Private input pipeline for bagging (reference autoensemble/common.py:
    # 43-56 _SecondaryTrainOpRunnerHook: the candidate trains on its own
    # batches; ensembles see its outputs on the shared batch).
    train_input_fn: Optional[Callable] = None
    train_iter: Optional[object] = None
    # per-batch cached outputs (detached)
    out_logits: Optional[torch.Tensor] = None
    out_last_layer: Optional[torch.Tensor] = None


@dataclasses.dataclass
class _EnsembleSpec:
    """One candidate ensemble (reference ensemble_builder.py:43-255)."""

    name: str
    candidate: Optional[EnsembleCandidate]
    ensemble: Optional[nn.Module]
    ensembler_name: str
    architecture: _Architecture
    optimizer: Optional[object] = None
    owner_rank: int = 0
    step: int = 0
    is_previous_best: bool = False
    # member resolution: list of ("frozen", name) | ("new", subnetwork name)
    members: Tuple = ()
    summary: Optional[_ScopedSummary] = None
    eval_loss: Optional[float] = None

    @property
    def new_subnetwork_names(self):
        return [m[1] for m in self.members if m[0] == "new"]


class _Iteration(object):
    """All candidates of one AdaNet round, training in lockstep steps."""

    def __init__(self, number: int, head, subnetwork_specs, ensemble_specs,
                 frozen_subnetworks: Dict[str, Subnetwork], train_manager,
                 max_iteration_steps: Optional[int], adanet_loss_decay: float,
                 device: torch.device, placement, use_streams: bool = True,
                 replicate_ensemble_in_training: bool = False,
                 to_device: Optional[Callable] = None,
                 use_graphs: bool = True,
                 frozen_logit_cache: Optional[_FrozenLogitCache] = None):
        self.number = number
        self.head = head
        self.subnetwork_specs: List[_SubnetworkSpec] = list(subnetwork_specs)
        self.ensemble_specs: List[_EnsembleSpec] = list(ensemble_specs)
        self.candidates = [
            _Candidate(spec, adanet_loss_decay) for spec in self.ensemble_specs
        ]
        self.frozen_subnetworks = dict(frozen_subnetworks)
        self.train_manager = train_manager
        self.max_iteration_steps = max_iteration_steps
        self.device = device
        self.placement = placement
        self.replicate_ensemble_in_training = replicate_ensemble_in_training
        self.to_device = to_device or (lambda f, l: (f, l))
        self.step = 0
        self._use_streams = use_streams and device.type == "cuda"
        if self._use_streams:
            for spec in self.subnetwork_specs:
                if spec.subnetwork is not None:
                    spec.stream = torch.cuda.Stream(device=device)
        # Device ring buffer of per-candidate adanet losses + per-subnetwork
        # losses: flushed to host EMAs every _LOSS_FLUSH_STEPS.
        n = len(self.ensemble_specs) + len(self.subnetwork_specs)
        self._loss_buf = torch.full((_LOSS_FLUSH_STEPS, max(n, 1)),
                                    float("nan"), device=device,
                                    dtype=torch.float32)
        # Device write cursor so the ring-buffer write is a pure device op
        # (required for hipGraph capture: the row index can't be a host
        # constant baked into the graph).
        self._loss_counter = torch.zeros((1,), device=device,
                                         dtype=torch.long)
        self._loss_buf_rows = 0
        self._row_recorded: List[set] = []  # names actually written per row
        self._nan_scalar = None
        self._frozen_event = None
        self.builder_hooks = []  # TrainOpSpec hooks collected by the engine
        # HBM frozen-logit cache: shared across iterations when the engine
        # passes its own (frozen member outputs are immutable forever).
        self._frozen_cache = (frozen_logit_cache
                              if frozen_logit_cache is not None
                              else _FrozenLogitCache())
        self._frozen_static = None
        self._need_last = None
        # hipGraph state
        self._use_graphs = use_graphs
        self._graph = None
        self._graph_sig = None
        self._graph_ok = None
        self._graph_warmups = 0
        self._graph_recorded: set = set()
        self._static_inputs = None

    # ------------------------------------------------------------------
    # training
    # ------------------------------------------------------------------

    @property
    def spec_names(self) -> List[str]:
        return ([s.name for s in self.subnetwork_specs] +
                [s.name for s in self.ensemble_specs])

    def compute_frozen_outputs(self, features, training: bool = False):
        """Runs every frozen previous-iteration subnetwork once per batch
        under no_grad; the resulting HBM-resident logits feed every
        candidate's fused mixer call (north star: frozen logits cached in
        288 GB HBM, never recomputed per candidate)."""
        outputs = {}
        with torch.no_grad():
            for name, sub in self.frozen_subnetworks.items():
                train_mode = training and self.replicate_ensemble_in_training
                sub.module.train(train_mode)
                last, logits = sub(features)
                outputs[name] = (last, logits)
        return outputs

    def _need_frozen_last_layer(self) -> bool:
        if self._need_last is None:
            need = False
            for spec in self.ensemble_specs:
                ens = spec.ensemble
                if ens is None:
                    continue
                if getattr(ens, "mixture_weights", "x") is None:
                    need = True  # MATRIX mixtures consume last_layer
                if getattr(ens, "add_mean_last_layer_predictions", False):
                    need = True
            self._need_last = need
        return self._need_last

    def _prepare_frozen(self, features, training: bool = False):
        """Frozen outputs for this batch, staged into STATIC buffers.

        The HBM logit cache (north star): inputs carrying an
        ``adanet_cache_key`` attribute (resident-dataset batches — e.g. the
        whole CIFAR-sized pool fits trivially in 288 GB) have their frozen
        members' outputs computed ONCE per iteration and replayed from HBM
        afterwards, so the per-step cost of the growing ensemble stays
        flat across epochs. Results land in static buffers so the
        hipGraph-captured step can consume them as plain graph inputs.
        """
        if not self.frozen_subnetworks:
            return {}
        key = getattr(features, "adanet_cache_key", None) if torch.is_tensor(
            features) else None
        cacheable = (key is not None
                     and not (training
                              and self.replicate_ensemble_in_training))
        keep_last = self._need_frozen_last_layer()
        if not cacheable:
            fresh = self.compute_frozen_outputs(features, training=training)
            outs = {n: (last if keep_last else None, logits)
                    for n, (last, logits) in fresh.items()}
            self._stage_frozen(outs)
            return self._frozen_static
        # Per-member cross-iteration cache: only members not yet seen on
        # this batch (i.e. the newly frozen winner) run a forward.
        outs = {}
        missing = []
        for name in self.frozen_subnetworks:
            ent = self._frozen_cache.get((key, name))
            if ent is None or (keep_last and ent[0] is None):
                missing.append(name)
            else:
                outs[name] = ent
        if missing:
            with torch.no_grad():
                for name in missing:
                    sub = self.frozen_subnetworks[name]
                    sub.module.eval()
                    last, logits = sub(features)
                    ent = (last.detach().clone() if keep_last else None,
                           logits.detach().clone())
                    self._frozen_cache.put((key, name), *ent)
                    outs[name] = ent
        self._stage_frozen(outs)
        return self._frozen_static

    def _stage_frozen(self, outs: Dict):
        """Copy per-member (last|None, logits) into the STATIC buffers a
        captured hipGraph reads; (re)allocates on first use / shape change."""
        if self._frozen_static is None or any(
                n not in self._frozen_static
                or self._frozen_static[n][1].shape != o[1].shape
                or (o[0] is not None and self._frozen_static[n][0] is None)
                for n, o in outs.items()):
            self._frozen_static = {
                n: [last.clone() if last is not None else None,
                    logits.clone()]
                for n, (last, logits) in outs.items()
            }
            # A captured graph reads the OLD static buffers — invalidate it
            # (batch-shape change mid-iteration).
            self._graph = None
            return
        srcs, dsts = [], []
        for name, (last_c, logits_c) in outs.items():
            last_s, logits_s = self._frozen_static[name]
            srcs.append(logits_c)
            dsts.append(logits_s)
            if last_c is not None and last_s is not None:
                srcs.append(last_c)
                dsts.append(last_s)
        if srcs and srcs[0].is_cuda and all(
                s.dtype == torch.bfloat16 and s.is_contiguous()
                and d.is_contiguous() for s, d in zip(srcs, dsts)):
            # One batched-copy launch instead of J-1 copyBuffer calls.
            from adanet_amd.ops import _extension
            _extension.require().multi_copy_bf16(srcs, dsts)
        else:
            for s, d in zip(srcs, dsts):
                d.copy_(s)

    def train_step(self, features, labels) -> None:
        """One lockstep training step for every still-active spec
        (the reference's single session.run over all candidate train ops,
        iteration.py:779-804 + hooks). The device work is hipGraph-captured
        after two warmup steps when eligible (single process, shared input,
        no LR schedule / dropout): the steady-state step then replays as
        one graph launch, eliminating the python launch-gap overhead that
        otherwise dominates (measured ~3.5 ms host vs ~1.1 ms GPU per step
        on the CIFAR DNN bench before graphing)."""
        tm = self.train_manager
        # Frozen members run (or replay from the HBM cache) OUTSIDE the
        # captured graph; the step consumes their static buffers.
        self._prepare_frozen(features, training=True)
        if self._graph_eligible():
            self._graphed_train_step(features, labels)
        else:
            losses_row = self._device_step(features, labels)
            self._write_loss_row(losses_row)
            self._finish_loss_row(set(losses_row.keys()))
        self._post_step_bookkeeping()

    def _post_step_bookkeeping(self):
        tm = self.train_manager
        for spec in self.subnetwork_specs:
            if spec.subnetwork is None:
                continue
            if tm.should_train(spec.name):
                if spec.optimizer is not None:
                    sched = getattr(spec.optimizer, "_adanet_lr_sched", None)
                    if sched is not None:
                        sched.step()
                spec.step += 1
                if (self.max_iteration_steps is not None
                        and spec.step >= self.max_iteration_steps):
                    tm.request_stop(spec.name, "Training is over.")
        for spec in self.ensemble_specs:
            if spec.ensemble is None:
                continue
            spec.step += 1
            if (self.max_iteration_steps is not None
                    and spec.step >= self.max_iteration_steps):
                tm.request_stop(spec.name, "Training is over.")
        self.step += 1
        if (self.max_iteration_steps is not None
                and self.step >= self.max_iteration_steps):
            for name in self.spec_names:
                tm.request_stop(name, "Training is over.")

    # ------------------------------------------------------------------
    # hipGraph capture of the steady-state step
    # ------------------------------------------------------------------

    def _graph_eligible(self) -> bool:
        if self._graph_ok is None:
            # Graphs need a collective-free step: single process, or any
            # world size under RoundRobin (candidates are rank-local; the
            # only collectives run at iteration end, outside the graph).
            no_step_collectives = (comm.world_size() <= 1
                                   or (self.placement is not None
                                       and not self.placement.data_parallel))
            ok = (self._use_graphs and self.device.type == "cuda"
                  and no_step_collectives)
            if ok:
                for spec in self.subnetwork_specs:
                    if spec.train_input_fn is not None:
                        ok = False
                    if spec.optimizer is not None and getattr(
                            spec.optimizer, "_adanet_lr_sched", None):
                        ok = False
                    # HipDropout is graph-safe: its seed is a device
                    # counter snapshot, so each replay draws a fresh mask.
            self._graph_ok = ok
        return bool(self._graph_ok)

    def _copy_static_inputs(self, features, labels):
        if self._static_inputs is not None:
            # Batch-shape change: rebuild statics and recapture.
            sf0, sl0 = self._static_inputs
            f0 = sf0[sorted(sf0)[0]] if isinstance(sf0, dict) else sf0
            f1 = (features[sorted(features)[0]]
                  if isinstance(features, dict) else features)
            if f0.shape != f1.shape or (
                    not isinstance(labels, dict)
                    and sl0.shape != labels.shape):
                self._static_inputs = None
                self._graph = None
        def _clone(x):
            return ({k: v.clone() for k, v in x.items()}
                    if isinstance(x, dict) else x.clone())

        def _copy(dst, src):
            if isinstance(dst, dict):
                for k in dst:
                    dst[k].copy_(src[k], non_blocking=True)
            else:
                dst.copy_(src, non_blocking=True)

        if self._static_inputs is None:
            self._static_inputs = (_clone(features), _clone(labels))
            return
        sf, sl = self._static_inputs
        _copy(sf, features)
        _copy(sl, labels)

    def _active_signature(self):
        return tuple(sorted(self.train_manager.stopped))

    def _graphed_train_step(self, features, labels):
        self._copy_static_inputs(features, labels)
        sf, sl = self._static_inputs
        sig = self._active_signature()
        if sig != self._graph_sig:
            self._graph = None
            self._graph_sig = sig
        if self._graph_warmups < 2 and self._graph is None:
            # Warmup on a side stream (allocator pool priming) — these ARE
            # real training steps.
            s = torch.cuda.Stream(device=self.device)
            s.wait_stream(torch.cuda.current_stream(self.device))
            with torch.cuda.stream(s):
                row = self._device_step(sf, sl)
                self._write_loss_row(row)
            torch.cuda.current_stream(self.device).wait_stream(s)
            self._graph_warmups += 1
            self._graph_recorded = set(row.keys())
            self._finish_loss_row(self._graph_recorded)
            return
        if self._graph is None:
            g = torch.cuda.CUDAGraph()
            try:
                with torch.cuda.graph(g):
                    row = self._device_step(sf, sl)
                    self._write_loss_row(row)
                self._graph = g
                self._graph_recorded = set(row.keys())
                log.info("hipGraph captured for iteration %s (%d specs)",
                         self.number, len(self._graph_recorded))
            except Exception as e:  # pragma: no cover - driver quirks
                log.warning("hipGraph capture failed (%r); eager fallback", e)
                self._use_graphs = False
                self._graph_ok = False
                row = self._device_step(sf, sl)
                self._write_loss_row(row)
                self._finish_loss_row(set(row.keys()))
                return
        self._graph.replay()
        self._finish_loss_row(self._graph_recorded)

    def _device_step(self, features, labels):
        """All of one step's device work; returns name -> loss tensor.
        Frozen outputs come from the static buffers prepared by
        train_step (outside any graph capture)."""
        tm = self.train_manager
        frozen_out = self._frozen_static or {}
        if self._use_streams:
            self._frozen_event = torch.cuda.Event()
            self._frozen_event.record()

        losses_row = {}
        # --- subnetworks: fwd + own head loss + fused optimizer ---
        for i, spec in enumerate(self.subnetwork_specs):
            if spec.subnetwork is None:
                continue  # not owned (round-robin)
            active = tm.should_train(spec.name)
            if self._use_streams:
                ctx = torch.cuda.stream(spec.stream)
                self._frozen_event.wait(spec.stream)
            else:
                ctx = _nullcontext()
            with ctx:
                spec.subnetwork.module.train(active)
                private_batch = None
                if active and spec.train_input_fn is not None:
                    if spec.train_iter is None:
                        spec.train_iter = iter(spec.train_input_fn())
                    try:
                        private_batch = next(spec.train_iter)
                    except StopIteration:
                        tm.request_stop(spec.name, "OutOfRange")
                        active = False
                if active:
                    if private_batch is not None:
                        pf, pl = private_batch
                        pf, pl = self.to_device(pf, pl)
                        last, logits = spec.subnetwork(pf)
                        loss = self._subnetwork_loss(spec, logits, pl, pf,
                                                     frozen_out)
                    else:
                        last, logits = spec.subnetwork(features)
                        loss = self._subnetwork_loss(spec, logits, labels,
                                                     features, frozen_out)
                    if spec.optimizer is not None:
                        spec.optimizer.zero_grad(set_to_none=True)
                        # Direct-to-arena dW/db writes (ops/linear.py):
                        # valid here because grads are only consumed via
                        # .grad (optimizer step / flat all-reduce).
                        with direct_grad_writes():
                            loss.backward()
                        if (self.placement is not None
                                and self.placement.data_parallel
                                and comm.world_size() > 1):
                            self._dp_allreduce(spec)
                        spec.optimizer.step()
                    losses_row[spec.name] = loss.detach()
                else:
                    with torch.no_grad():
                        spec.subnetwork.module.eval()
                        last, logits = spec.subnetwork(features)
                if private_batch is not None:
                    # Ensembles consume outputs on the SHARED batch
                    # (reference common.py:146-180: model_fn invoked twice
                    # when bagging).
                    with torch.no_grad():
                        was_training = spec.subnetwork.module.training
                        spec.subnetwork.module.eval()
                        last, logits = spec.subnetwork(features)
                        spec.subnetwork.module.train(was_training)
                spec.out_logits = logits.detach()
                spec.out_last_layer = last.detach()

        # --- candidate ensembles: mixture weights on the AdaNet objective ---
        for spec in self.ensemble_specs:
            if spec.ensemble is None:
                continue  # not owned
            if self._use_streams:
                stream = self._member_stream(spec)
                ctx = torch.cuda.stream(stream) if stream else _nullcontext()
            else:
                ctx = _nullcontext()
            with ctx:
                loss_t = self._ensemble_adanet_loss(spec, frozen_out, labels,
                                                    train=True)
                if loss_t is not None:
                    losses_row[spec.name] = loss_t.detach()

        if self._use_streams:
            # Device-side join: the default stream waits on every candidate
            # stream (no host synchronization in the steady-state step).
            cur = torch.cuda.current_stream(self.device)
            for spec in self.subnetwork_specs:
                if spec.stream is not None:
                    cur.wait_stream(spec.stream)
        return losses_row

    def _dp_allreduce(self, spec):
        """Gradient all-reduce for one candidate (ReplicationStrategy).

        On GPU the collective runs on a dedicated comm stream gated by an
        event on the candidate's compute stream, so the xGMI ring of
        candidate i overlaps candidate i+1's forward/backward enqueued
        right after (the round-1 structure reduced strictly AFTER backward
        on the compute stream — zero overlap). All ranks enqueue
        collectives in identical spec order, so RCCL ordering holds.
        """
        arenas = getattr(spec.optimizer, "_arenas", None)
        bufs = (spec.optimizer.flat_grad_buffers() if arenas else None)
        if self.device.type == "cuda":
            if not hasattr(self, "_comm_stream") or self._comm_stream is None:
                self._comm_stream = torch.cuda.Stream(device=self.device)
            cur = torch.cuda.current_stream(self.device)
            ev = torch.cuda.Event()
            ev.record(cur)
            self._comm_stream.wait_event(ev)
            with torch.cuda.stream(self._comm_stream):
                if bufs is not None:
                    comm.allreduce_buffers(bufs)
                else:
                    comm.allreduce_gradients(
                        list(spec.subnetwork.module.parameters()))
            # optimizer.step (enqueued next on `cur`) consumes the reduced
            # grads: device-side join, no host sync.
            cur.wait_stream(self._comm_stream)
        else:
            if bufs is not None:
                comm.allreduce_buffers(bufs)
            else:
                comm.allreduce_gradients(
                    list(spec.subnetwork.module.parameters()))

    def _subnetwork_loss(self, spec, logits, labels, features, frozen_out):
        """Head loss, or the builder's custom loss hook when provided
        (enables knowledge distillation: the improve_nas search space adds
        a distillation term against the previous ensemble's logits —
        reference research/improve_nas/trainer/improve_nas.py:41-59,
        166-181)."""
        hook = getattr(spec.builder, "build_subnetwork_loss", None)
        if hook is None:
            return self.head.loss(logits, labels)

        def frozen_outputs():
            if spec.train_input_fn is not None:
                # Private (bagging) batch: the shared-batch frozen cache
                # doesn't apply — run the frozen members on this batch.
                return self.compute_frozen_outputs(features)
            return frozen_out

        def prev_ensemble_logits():
            prev = self.previous_best_spec
            if prev is None or prev.ensemble is None:
                return None
            sub_logits, sub_last = self._gather_member_outputs(
                prev, frozen_outputs())
            if sub_logits is None:
                return None
            with torch.no_grad():
                return prev.ensemble.logits_from(sub_logits, sub_last)

        return hook(head=self.head, logits=logits, labels=labels,
                    features=features,
                    previous_ensemble_logits_fn=prev_ensemble_logits,
                    frozen_outputs_fn=frozen_outputs)

    @property
    def previous_best_spec(self) -> Optional[_EnsembleSpec]:
        for spec in self.ensemble_specs:
            if spec.is_previous_best:
                return spec
        return None

    def _member_stream(self, spec: _EnsembleSpec):
        for kind, name in spec.members:
            if kind == "new":
                for s in self.subnetwork_specs:
                    if s.name == name and s.stream is not None:
                        return s.stream
        return None

    def _gather_member_outputs(self, spec: _EnsembleSpec, frozen_out):
        sub_logits, sub_last = [], []
        for kind, name in spec.members:
            if kind == "frozen":
                last, logits = frozen_out[name]
            else:
                s = self._subnetwork_spec(name)
                logits, last = s.out_logits, s.out_last_layer
                if logits is None:
                    return None, None
            sub_logits.append(logits)
            sub_last.append(last)
        return sub_logits, sub_last

    def _subnetwork_spec(self, name: str) -> _SubnetworkSpec:
        """Lookup by spec name or by the builder's (candidate member) name."""
        for s in self.subnetwork_specs:
            if s.name == name or s.builder.name == name:
                return s
        raise KeyError(name)

    def _ensemble_adanet_loss(self, spec: _EnsembleSpec, frozen_out, labels,
                              train: bool):
        """adanet_loss = head loss + complexity regularization
        (reference ensemble_builder.py:420-426)."""
        sub_logits, sub_last = self._gather_member_outputs(spec, frozen_out)
        if sub_logits is None:
            return None
        train_mixture = (train and spec.optimizer is not None
                         and not spec.is_previous_best
                         and self.train_manager.should_train(spec.name))
        if train_mixture:
            logits = spec.ensemble.logits_from(sub_logits, sub_last)
            loss = self.head.loss(logits, labels)
            creg = spec.ensemble.complexity_regularization()
            adanet_loss = loss + creg
            spec.optimizer.zero_grad(set_to_none=True)
            # mixer dW/db write direct-to-arena (ops/mixer.py backward);
            # the L1-penalty grad still flows through AccumulateGrad and
            # ADDS into the same views.
            with direct_grad_writes():
                adanet_loss.backward()
            spec.optimizer.step()
            return adanet_loss
        with torch.no_grad():
            logits = spec.ensemble.logits_from(sub_logits, sub_last)
            loss = self.head.loss(logits, labels)
            creg = spec.ensemble.complexity_regularization()
            return loss + creg

    def _ensemble_eval_metric(self, spec: _EnsembleSpec, frozen_out, labels,
                              metric_name: str):
        """A head metric of the candidate ensemble on one batch (custom
        Evaluator metric_name, reference evaluator.py:31-60: any metric in
        the candidate's eval dict may drive selection)."""
        sub_logits, sub_last = self._gather_member_outputs(spec, frozen_out)
        if sub_logits is None:
            return None
        with torch.no_grad():
            logits = spec.ensemble.logits_from(sub_logits, sub_last)
            metrics = self.head.metrics(logits, labels)
        if metric_name not in metrics:
            raise ValueError(
                "Evaluator metric_name %r not produced by the head "
                "(available: %s)" % (metric_name, sorted(metrics)))
        return float(metrics[metric_name])

    # ------------------------------------------------------------------
    # loss ring buffer -> host EMAs
    # ------------------------------------------------------------------

    def _write_loss_row(self, losses_row: Dict[str, torch.Tensor]):
        """Pure device ops: scatter this step's losses into the ring buffer
        at the device write cursor (hipGraph-capturable). One stack kernel
        builds the row (not a per-spec D2D copy each)."""
        names = self.spec_names
        if self._nan_scalar is None:
            self._nan_scalar = torch.full((), float("nan"),
                                          device=self.device,
                                          dtype=torch.float32)
        vec = torch.stack([
            losses_row.get(name, self._nan_scalar).float().reshape(())
            for name in names
        ])
        idx = torch.remainder(self._loss_counter, _LOSS_FLUSH_STEPS)
        self._loss_buf.index_copy_(0, idx, vec.unsqueeze(0))
        self._loss_counter.add_(1)

    def _finish_loss_row(self, recorded: set):
        """Host bookkeeping for one written row + periodic flush."""
        self._row_recorded.append(recorded)
        self._loss_buf_rows += 1
        if self._loss_buf_rows >= _LOSS_FLUSH_STEPS:
            self.flush_losses()

    def flush_losses(self):
        """Reads the device loss buffer once and updates host-side EMAs /
        NaN bookkeeping (reference _NanLossHook warns on NaN,
        iteration.py:121-147; EMA in candidate.py:117-129). NaN for a
        RECORDED spec means genuine divergence (the buffer's un-recorded
        slots are also NaN but tracked separately)."""
        if self._loss_buf_rows == 0:
            return
        rows = self._loss_buf[:self._loss_buf_rows].cpu().numpy()
        n_sub = len(self.subnetwork_specs)
        base_step = self.step - self._loss_buf_rows
        for r in range(rows.shape[0]):
            recorded = self._row_recorded[r]
            for i, spec in enumerate(self.subnetwork_specs):
                if spec.name not in recorded:
                    continue
                v = float(rows[r, i])
                spec.last_loss = v
                if math.isnan(v):
                    log.warning("'%s' diverged with loss = NaN.", spec.name)
            for j, cand in enumerate(self.candidates):
                if cand.ensemble_spec.name not in recorded:
                    continue
                # NaN while training poisons the EMA so selection surfaces
                # divergence (reference iteration.py:1040-1046).
                cand.update(float(rows[r, n_sub + j]))
        # TensorBoard-style per-candidate charts (the reference logs `loss`
        # and `adanet_loss` under each candidate's scope,
        # summary.py:262-296): one event per flush to keep IO off the step.
        step = self.step
        for i, spec in enumerate(self.subnetwork_specs):
            if spec.summary is not None and not math.isnan(spec.last_loss):
                spec.summary.set_step(step)
                spec.summary.scalar("loss", spec.last_loss)
        for spec, cand in zip(self.ensemble_specs, self.candidates):
            if spec.summary is not None and cand.adanet_loss not in (
                    float("inf"),):
                spec.summary.set_step(step)
                spec.summary.scalar("adanet_loss", cand.adanet_loss)
        self._loss_buf.fill_(float("nan"))
        self._loss_counter.fill_(0)
        self._loss_buf_rows = 0
        self._row_recorded = []

    def is_over(self) -> bool:
        return self.train_manager.is_over(self.spec_names)

    # ------------------------------------------------------------------
    # selection
    # ------------------------------------------------------------------

    def adanet_losses(self) -> List[float]:
        """Per-candidate EMA adanet losses, all-gathered across owners in
        round-robin placement."""
        self.flush_losses()
        local = {}
        for i, cand in enumerate(self.candidates):
            if cand.ensemble_spec.ensemble is not None:
                local[i] = cand.adanet_loss
        gathered = comm.all_gather_objects(local)
        merged = {}
        for d in gathered:
            merged.update(d)
        return [
            merged.get(i, float("inf")) for i in range(len(self.candidates))
        ]

    def best_candidate_index(self,
                             override: Optional[int] = None,
                             losses: Optional[Sequence[float]] = None) -> int:
        """np.nanargmin over adanet losses: diverged (NaN) candidates LOSE
        (reference bookkeeping selection, estimator.py:1494-1512 uses
        np.nanargmin — distinct from the in-graph prediction mux's NaN->-inf
        at iteration.py:1040-1046). All-NaN raises like np.nanargmin."""
        if override is not None:
            return int(override)
        if len(self.candidates) == 1:
            return 0
        vals = list(losses) if losses is not None else self.adanet_losses()
        best, best_i = None, 0
        for i, v in enumerate(vals):
            if math.isnan(v):
                continue
            if best is None or v < best:
                best, best_i = v, i
        if best is None:
            from adanet_amd.core.estimator import NanLossDuringTrainingError
            raise NanLossDuringTrainingError(
                "Iteration {}: every candidate's adanet_loss is NaN "
                "(training diverged).".format(self.number))
        return best_i

    # ------------------------------------------------------------------
    # evaluation
    # ------------------------------------------------------------------

    def evaluate_candidates(self, input_iter, steps: Optional[int],
                            to_device: Callable,
                            metric_name: str = "adanet_loss") -> List[float]:
        """Mean eval metric per candidate over shared eval batches
        (reference evaluator.py:97-140: same batches for all candidates).
        Default metric is the fused adanet_loss; any head metric (e.g.
        "accuracy" with a MAXIMIZE objective) may drive selection.
        Round-robin: each rank evaluates the candidates it owns; results
        are merged by adanet_losses()-style all-gather in the caller."""
        # Accumulate per-candidate losses as DEVICE tensors: one host sync
        # per candidate at the end instead of candidates x eval-batches
        # blocking .item() round-trips mid-phase.
        sums = [0.0] * len(self.ensemble_specs)
        count = 0
        step = 0
        while steps is None or step < steps:
            try:
                features, labels = next(input_iter)
            except StopIteration:
                break
            features, labels = to_device(features, labels)
            frozen_out = self._prepare_frozen(features)
            with torch.no_grad():
                for spec in self.subnetwork_specs:
                    if spec.subnetwork is None:
                        continue
                    spec.subnetwork.module.eval()
                    last, logits = spec.subnetwork(features)
                    spec.out_logits, spec.out_last_layer = logits, last
                for i, spec in enumerate(self.ensemble_specs):
                    if spec.ensemble is None:
                        continue
                    if metric_name == "adanet_loss":
                        loss = self._ensemble_adanet_loss(
                            spec, frozen_out, labels, train=False)
                    else:
                        loss = self._ensemble_eval_metric(
                            spec, frozen_out, labels, metric_name)
                    if loss is not None:
                        sums[i] = sums[i] + (loss.detach() if isinstance(
                            loss, torch.Tensor) else loss)
            count += 1
            step += 1
        if count == 0:
            return [float("nan")] * len(self.ensemble_specs)
        out = []
        for i, spec in enumerate(self.ensemble_specs):
            if spec.ensemble is None:
                out.append(float("nan"))
            else:
                spec.eval_loss = float(sums[i]) / count
                out.append(spec.eval_loss)
        return out


class _nullcontext(object):

    def __enter__(self):
        return None

    def __exit__(self, *a):
        return False
