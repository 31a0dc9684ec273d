"""Head unit tests (reference: tf.estimator head contracts consumed at
adanet/core/ensemble_builder.py:571-583; multi-head estimator_test.py:1517)."""

import pytest
import torch

from adanet_amd.head import (BinaryClassHead, MultiClassHead, MultiHead,
                             RegressionHead)


def test_multiclass_loss_matches_torch_ce():
    h = MultiClassHead(10, label_smoothing=0.1)
    torch.manual_seed(0)
    logits = torch.randn(64, 10)
    labels = torch.randint(0, 10, (64,))
    want = torch.nn.functional.cross_entropy(logits, labels,
                                             label_smoothing=0.1)
    got = h.loss(logits, labels)
    assert abs(float(got) - float(want)) < 1e-4
    preds = h.predictions(logits)
    assert preds["class_ids"].shape == (64,)
    assert torch.allclose(preds["probabilities"].sum(dim=1),
                          torch.ones(64), atol=1e-5)
    m = h.metrics(logits, labels)
    assert 0.0 <= m["accuracy"] <= 1.0 and "average_loss" in m


def test_multiclass_validates_classes():
    with pytest.raises(ValueError):
        MultiClassHead(1)


def test_multihead_sums_losses_and_namespaces():
    h = MultiHead({"a": MultiClassHead(3), "b": RegressionHead(2)})
    assert h.logits_dimension == 5
    torch.manual_seed(1)
    logits = torch.randn(16, 5)
    labels = {"a": torch.randint(0, 3, (16,)), "b": torch.randn(16, 2)}
    la = MultiClassHead(3).loss(logits[:, :3], labels["a"])
    lb = RegressionHead(2).loss(logits[:, 3:], labels["b"])
    assert abs(float(h.loss(logits, labels)) - float(la + lb)) < 1e-5
    preds = h.predictions(logits)
    assert "a/class_ids" in preds and "b/predictions" in preds
    m = h.metrics(logits, labels)
    assert "a/accuracy" in m and "average_loss" in m


def test_regression_head_mse():
    h = RegressionHead(1)
    logits = torch.tensor([[1.0], [2.0]])
    labels = torch.tensor([[0.0], [4.0]])
    assert abs(float(h.loss(logits, labels)) - 2.5) < 1e-6


def test_binary_head_contract():
    h = BinaryClassHead()
    assert h.logits_dimension == 1
    torch.manual_seed(2)
    logits = torch.randn(32, 1)
    labels = (torch.rand(32) > 0.5).long()
    want = torch.nn.functional.binary_cross_entropy_with_logits(
        logits.reshape(-1), labels.float())
    assert abs(float(h.loss(logits, labels)) - float(want)) < 1e-5
    p = h.predictions(logits)
    assert set(p) == {"logits", "probabilities", "class_ids"}


def test_train_hooks_fire_and_stop():
    from adanet_amd.hooks import EveryNSteps, StopAfterSteps
    seen = []
    h1 = EveryNSteps(2, lambda step: seen.append(step))
    h2 = StopAfterSteps(5)
    for s in range(1, 8):
        h1.after_step(s)
        h2.after_step(s)
    assert seen == [2, 4, 6]
    assert h2.should_stop


def test_comm_singleprocess_fallbacks():
    """comm helpers degrade to no-ops without a process group."""
    from adanet_amd.distributed import comm
    assert not comm.is_initialized()
    assert comm.world_size() == 1 and comm.rank() == 0 and comm.is_chief()
    assert comm.broadcast_object({"x": 1}) == {"x": 1}
    assert comm.all_gather_objects("v") == ["v"]
    comm.barrier()  # no-op
