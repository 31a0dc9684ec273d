"""Serving loader + eval metric stores."""

import math
import os

import pytest
import torch

import adanet_amd
from adanet_amd import serving
from adanet_amd.core.eval_metrics import (_EnsembleMetrics,
                                          _EvalMetricsStore,
                                          _IterationMetrics)
from adanet_amd.head import MultiClassHead
from adanet_amd.models import simple_dnn


def test_eval_metrics_store_streaming_mean():
    s = _EvalMetricsStore()
    s.update({"acc": 1.0}, n=2)
    s.update({"acc": 0.0}, n=2)
    assert s.result()["acc"] == 0.5


def test_iteration_metrics_best_mux():
    e0 = _EnsembleMetrics("archA")
    e0.update({"loss": 1.0})
    e1 = _EnsembleMetrics("archB")
    e1.update({"loss": 0.5})
    im = _IterationMetrics(2, [e0, e1], replay_indices=[0, 1, 1])
    best = im.best_eval_metrics(1)
    assert best["loss"] == 0.5
    assert best["architecture/adanet/ensembles"] == "archB"
    assert best["iteration"] == 2
    assert best["best_ensemble_index_2"] == 1


def test_export_and_serving_roundtrip(tmp_path, synthetic_classification):
    X, Y, input_fn = synthetic_classification
    gen = simple_dnn.Generator(layer_size=8)
    est = adanet_amd.Estimator(
        head=MultiClassHead(4), subnetwork_generator=gen,
        max_iteration_steps=10, model_dir=str(tmp_path / "m"),
        config=adanet_amd.RunConfig(tf_random_seed=42))
    est.train(input_fn, max_steps=20)
    export_dir = est.export_saved_model(str(tmp_path / "export"))

    servable = serving.load_ensemble(
        export_dir, subnetwork_generator=simple_dnn.Generator(layer_size=8),
        head=MultiClassHead(4), device="cpu")
    with torch.no_grad():
        out = servable(X[:8])
    assert out.shape == (8, 4)
    # Must agree with the live estimator's frozen-best forward.
    live, _ = est._load_frozen_best()
    with torch.no_grad():
        ref = live(X[:8])
    assert torch.allclose(out.float(), ref.float(), atol=1e-4)
    preds = servable.predict(X[:4])
    assert "probabilities" in preds


def test_torchscript_export_standalone(tmp_path, synthetic_classification):
    """Traced frozen ensemble: loadable with plain torch.jit.load, outputs
    match the live ensemble."""
    X, Y, input_fn = synthetic_classification
    gen = simple_dnn.Generator(layer_size=8)
    est = adanet_amd.Estimator(
        head=MultiClassHead(4), subnetwork_generator=gen,
        max_iteration_steps=10, model_dir=str(tmp_path / "m"),
        config=adanet_amd.RunConfig(tf_random_seed=42))
    est.train(input_fn, max_steps=20)
    path = str(tmp_path / "ensemble_ts.pt")
    serving.export_torchscript(est, X[:8], path)
    loaded = torch.jit.load(path)
    with torch.no_grad():
        out = loaded(X[:8].float())
    live, _ = est._load_frozen_best()
    with torch.no_grad():
        ref = live(X[:8])
    assert out.shape == ref.shape
    # live path rounds through bf16; the portable artifact is full fp32.
    assert torch.allclose(out, ref.float(), atol=1e-2), (
        (out - ref.float()).abs().max())


def test_auc_accumulator_vs_sklearn():
    import torch
    from sklearn.metrics import roc_auc_score
    from adanet_amd.core.eval_metrics import AUCAccumulator
    torch.manual_seed(0)
    acc = AUCAccumulator(num_thresholds=400)
    all_s, all_y = [], []
    for _ in range(5):  # streaming over batches
        y = (torch.rand(512) > 0.6).long()
        s = (0.3 * torch.randn(512) + 0.3 + 0.35 * y.float()).clamp(0, 1)
        acc.update(s, y)
        all_s.append(s)
        all_y.append(y)
    got = acc.value()
    want = roc_auc_score(torch.cat(all_y).numpy(), torch.cat(all_s).numpy())
    assert abs(got["auc"] - want) < 0.01, (got["auc"], want)
    assert 0.0 <= got["precision"] <= 1.0
    assert 0.0 <= got["recall"] <= 1.0


def test_binary_head_metrics_include_auc():
    import torch
    from adanet_amd.head import BinaryClassHead
    torch.manual_seed(1)
    h = BinaryClassHead()
    logits = torch.randn(256, 1)
    labels = (torch.rand(256) > 0.5).long()
    m = h.metrics(logits, labels)
    for k in ("accuracy", "average_loss", "auc", "precision", "recall"):
        assert k in m


def test_export_program_roundtrip(tmp_path, synthetic_classification):
    """torch.export artifact: loadable with torch.export.load, dynamic
    batch honored, outputs match the live ensemble."""
    X, Y, input_fn = synthetic_classification
    gen = simple_dnn.Generator(layer_size=8)
    est = adanet_amd.Estimator(
        head=MultiClassHead(4), subnetwork_generator=gen,
        max_iteration_steps=10, model_dir=str(tmp_path / "m"),
        config=adanet_amd.RunConfig(tf_random_seed=42))
    est.train(input_fn, max_steps=20)
    path = str(tmp_path / "ensemble.pt2")
    serving.export_program(est, X[:8], path)
    ep = torch.export.load(path)
    with torch.no_grad():
        out = ep.module()(X[:8].float())
        out16 = ep.module()(X[:16].float())  # dynamic batch
    live, _ = est._load_frozen_best()
    with torch.no_grad():
        ref = live(X[:8])
    assert out.shape == ref.shape and out16.shape[0] == 16
    assert torch.allclose(out, ref.float(), atol=1e-2)
