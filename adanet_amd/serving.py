"""Serving: load an exported AdaNet ensemble for inference.

The analog of reference SavedModel export/serving
(adanet/core/estimator.py:1090-1146 + iteration.py:1111-1186 export
muxing): export_saved_model() writes a self-contained saved_model.pt
bundle; load_ensemble() reconstructs the frozen best ensemble given the
same (deterministic) subnetwork generator.
"""

from __future__ import annotations

import os
from typing import Optional

import torch


def to_portable_module(module: torch.nn.Module) -> torch.nn.Module:
    """Converts adanet_amd device modules (HipLinear/HipLayerNorm/
    HipDropout) into pure-torch equivalents with identical weights, so the
    frozen ensemble can be TorchScript-traced and served WITHOUT
    adanet_amd or a GPU (the SavedModel-portability analog: reference
    exports are runnable under stock TF serving)."""
    import copy

    from torch import nn

    from adanet_amd.ops.dropout import HipDropout
    from adanet_amd.ops.layernorm import HipLayerNorm
    from adanet_amd.ops.linear import HipLinear

    seen = []

    def convert(m):
        seen.append(m)
        # Member subnetworks hang off `subnetwork` dataclass attributes
        # (WeightedSubnetwork / MeanEnsemble), invisible to named_children.
        sub = getattr(m, "subnetwork", None)
        if sub is not None and isinstance(getattr(sub, "module", None),
                                          nn.Module):
            convert(sub.module)
        subs = getattr(m, "_subnetworks", None)
        if isinstance(subs, (list, tuple)):
            for s in subs:
                if isinstance(getattr(s, "module", None), nn.Module):
                    convert(s.module)
        for name, child in list(m.named_children()):
            if isinstance(child, HipLinear):
                lin = nn.Linear(child.in_features, child.out_features)
                with torch.no_grad():
                    lin.weight.copy_(
                        child.weight[:child.out_features].float())
                    if child.bias is not None:
                        lin.bias.copy_(
                            child.bias[:child.out_features].float())
                    else:
                        lin.bias.zero_()
                repl = [nn.Flatten(start_dim=1), lin]
                if child.activation == "relu":
                    repl.append(nn.ReLU())
                setattr(m, name, nn.Sequential(*repl))
            elif isinstance(child, HipLayerNorm):
                ln = nn.LayerNorm(child.dim, eps=child.eps,
                                  elementwise_affine=child.weight is not None)
                with torch.no_grad():
                    if child.weight is not None:
                        ln.weight.copy_(child.weight.float())
                        ln.bias.copy_(child.bias.float())
                setattr(m, name, ln)
            elif isinstance(child, HipDropout):
                setattr(m, name, nn.Dropout(child.p))
            else:
                convert(child)

    portable = copy.deepcopy(module).cpu().float()
    convert(portable)
    # Dataclass-held member modules are free tensors to torch.jit.trace;
    # inference artifact -> no grads anywhere.
    for m in seen:
        for p in m.parameters():
            p.requires_grad_(False)
    portable.eval()
    return portable


class _TracedEnsembleWrapper(torch.nn.Module):
    """Adapter so tracing sees a plain logits forward."""

    def __init__(self, ensemble):
        super().__init__()
        self.ensemble = ensemble

    def forward(self, features):
        return self.ensemble(features)


def export_torchscript(estimator, example_features, path: str):
    """Traces the frozen best ensemble into a standalone TorchScript file
    loadable with plain `torch.jit.load` (no adanet_amd dependency).
    """
    t = estimator._iteration_number
    built, _ = estimator._rebuild_previous_ensemble(t, example_features)
    if built is None:
        raise ValueError("No trained ensemble to export — train() first.")
    portable = to_portable_module(built)
    wrapper = _TracedEnsembleWrapper(portable)
    ex = example_features
    if torch.is_tensor(ex):
        ex = ex.detach().cpu().float()
    with torch.no_grad():
        traced = torch.jit.trace(wrapper, ex)
    traced.save(path)
    return path


def export_program(estimator, example_features, path: str,
                   dynamic_batch: bool = True):
    """torch.export artifact of the frozen best ensemble (.pt2).

    Unlike the TorchScript trace, the ExportedProgram records a
    functionalized ATen graph with explicit input shapes (batch
    symbolically dynamic by default) — loadable with `torch.export.load`
    in any adanet_amd-free environment, and the graph is the stable
    serving/compile interface (the reference's SavedModel analog;
    estimator.py export_saved_model_for_serving).
    """
    t = estimator._iteration_number
    built, _ = estimator._rebuild_previous_ensemble(t, example_features)
    if built is None:
        raise ValueError("No trained ensemble to export — train() first.")
    portable = to_portable_module(built)
    wrapper = _TracedEnsembleWrapper(portable)
    ex = example_features
    if torch.is_tensor(ex):
        ex = ex.detach().cpu().float()
    dynamic_shapes = None
    if dynamic_batch and torch.is_tensor(ex):
        batch = torch.export.Dim("batch", min=1)
        dynamic_shapes = ((batch,) + (None,) * (ex.dim() - 1),)
    ep = torch.export.export(wrapper, (ex,), dynamic_shapes=dynamic_shapes)
    torch.export.save(ep, path)
    return path


class ServableEnsemble(torch.nn.Module):
    """Frozen best ensemble + prediction helpers."""

    def __init__(self, estimator, head):
        super().__init__()
        self._est = estimator
        self._head = head
        self._ens = None

    def forward(self, features):
        ens, _ = self._est._load_frozen_best()
        return ens(features)

    def predict(self, features):
        with torch.no_grad():
            return self._head.predictions(self(features))


def load_ensemble(export_dir: str, subnetwork_generator, head,
                  device: Optional[str] = None) -> ServableEnsemble:
    """Loads a saved_model.pt bundle produced by export_saved_model().

    Args:
        export_dir: the timestamped export directory.
        subnetwork_generator: the SAME (deterministic) generator used for
            training — builders are re-invoked to reconstruct module
            structure, then weights load from the bundle (the reference
            replays architectures the same way, estimator.py:1785-1882).
        head: the head used in training.
        device: target device (default: cuda if available).
    """
    from adanet_amd.config import RunConfig
    from adanet_amd.core.estimator import Estimator

    path = os.path.join(export_dir, "saved_model.pt")
    payload = torch.load(path, map_location="cpu", weights_only=False)
    if payload.get("format") != "adanet_amd.v1":
        raise ValueError("Unknown export format in %s" % path)
    est = Estimator.__new__(Estimator)
    # Minimal state needed by _load_frozen_best/_rebuild_previous_ensemble.
    est._head = head
    est._subnetwork_generator = subnetwork_generator
    est._report_materializer = None
    est._ensemblers = []
    from adanet_amd.ensemble import ComplexityRegularizedEnsembler
    est._ensemblers = [ComplexityRegularizedEnsembler()]
    est._config = RunConfig(device=device)
    est._device = est._config.resolve_device()
    est._architectures = {
        int(k): v for k, v in payload["architectures"].items()
    }
    est._frozen_states = payload["frozen_states"]
    est._best_ensemble_state = payload["ensemble_state"]
    est._replay_indices = list(payload.get("replay_indices", []))
    est._iteration_number = max(est._architectures) + 1 if (
        est._architectures) else 0
    est._global_step = 0
    est._frozen_best_cache = None
    return ServableEnsemble(est, head)
