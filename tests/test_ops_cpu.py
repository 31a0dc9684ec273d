"""CPU reference-path tests for adanet_amd.ops — these same ops are compared
against the HIP kernels in test_gpu_kernels.py (marked gpu)."""

import math

import pytest
import torch

from adanet_amd.ops import (FusedAdam, FusedSGD, HipDropout, HipLayerNorm,
                            HipLinear, softmax_xent, weighted_sum_logits)
from adanet_amd.ops.optim import CosineLR, make_optimizer


def test_hip_linear_matches_torch_linear_fp32():
    torch.manual_seed(0)
    lin = HipLinear(16, 8, dtype=torch.float32)
    ref = torch.nn.Linear(16, 8)
    with torch.no_grad():
        ref.weight.copy_(lin.weight[:8])
        ref.bias.copy_(lin.bias[:8])
    x = torch.randn(4, 16)
    out = lin(x)
    expected = ref(x)
    assert torch.allclose(out, expected, atol=1e-5)


def test_hip_linear_relu_and_grads():
    torch.manual_seed(0)
    lin = HipLinear(8, 6, activation="relu", dtype=torch.float32)
    x = torch.randn(5, 8, requires_grad=True)
    y = lin(x)
    assert (y >= 0).all()
    y.sum().backward()
    assert lin.weight.grad is not None
    assert lin.bias.grad is not None
    assert x.grad is not None
    # ReLU mask: zero rows of grad where output was clamped.
    ref = torch.relu(x.detach() @ lin.weight[:6].t().detach() +
                     lin.bias[:6].detach())
    assert torch.allclose(y, ref, atol=1e-5)


def test_hip_linear_padding():
    lin = HipLinear(8, 10, dtype=torch.float32)  # 10 -> padded 32
    assert lin.padded_out == 32
    assert lin.weight.shape == (32, 8)
    assert (lin.weight[10:] == 0).all()
    y = lin(torch.randn(3, 8))
    assert y.shape == (3, 10)


def test_weighted_sum_logits_flat_tensor():
    a = torch.full((4, 3), 1.0)
    b = torch.full((4, 3), 3.0)
    flat = torch.tensor([0.5, 0.25])
    out = weighted_sum_logits([a, b], flat)
    assert torch.allclose(out, torch.full((4, 3), 1.25))
    flat_v = torch.tensor([[1.0, 0.0, 1.0], [0.0, 1.0, 0.0]])
    out_v = weighted_sum_logits([a, b], flat_v)
    assert torch.allclose(out_v, torch.tensor([1.0, 3.0, 1.0]).expand(4, 3))


def test_softmax_xent_matches_cross_entropy():
    torch.manual_seed(0)
    logits = torch.randn(32, 10)
    labels = torch.randint(0, 10, (32,))
    ours = softmax_xent(logits, labels)
    ref = torch.nn.functional.cross_entropy(logits, labels)
    assert torch.allclose(ours, ref, atol=1e-6)
    ours_ls = softmax_xent(logits, labels, label_smoothing=0.1)
    ref_ls = torch.nn.functional.cross_entropy(logits, labels,
                                               label_smoothing=0.1)
    assert torch.allclose(ours_ls, ref_ls, atol=1e-6)


def test_weighted_sum_logits_cpu():
    a = torch.full((4, 3), 1.0)
    b = torch.full((4, 3), 3.0)
    w1 = torch.tensor(0.5)
    w2 = torch.tensor(0.25)
    bias = torch.tensor([0.1, 0.1, 0.1])
    out = weighted_sum_logits([a, b], [w1, w2], bias)
    assert torch.allclose(out, torch.full((4, 3), 0.5 + 0.75 + 0.1))


def test_fused_sgd_matches_torch_sgd_fp32():
    torch.manual_seed(0)
    p1 = torch.nn.Parameter(torch.randn(10))
    p2 = torch.nn.Parameter(p1.detach().clone())
    o1 = FusedSGD([p1], lr=0.1, momentum=0.9, weight_decay=0.01)
    o2 = torch.optim.SGD([p2], lr=0.1, momentum=0.9, weight_decay=0.01)
    for _ in range(5):
        g = torch.randn(10)
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
    assert torch.allclose(p1, p2, atol=1e-5)


def test_fused_adam_matches_torch_adam_fp32():
    torch.manual_seed(0)
    p1 = torch.nn.Parameter(torch.randn(10))
    p2 = torch.nn.Parameter(p1.detach().clone())
    o1 = FusedAdam([p1], lr=0.01)
    o2 = torch.optim.Adam([p2], lr=0.01)
    for _ in range(5):
        g = torch.randn(10)
        p1.grad = g.clone()
        p2.grad = g.clone()
        o1.step()
        o2.step()
    assert torch.allclose(p1, p2, atol=1e-5)


def test_optimizer_registry():
    p = [torch.nn.Parameter(torch.ones(2))]
    assert isinstance(make_optimizer("sgd", p, lr=0.1), FusedSGD)
    assert isinstance(make_optimizer("momentum", p, lr=0.1), FusedSGD)
    assert isinstance(make_optimizer("adam", p, lr=0.1), FusedAdam)
    with pytest.raises(ValueError):
        make_optimizer("bogus", p, lr=0.1)


def test_cosine_lr_decays_to_zero():
    p = [torch.nn.Parameter(torch.ones(2))]
    opt = FusedSGD(p, lr=1.0)
    sched = CosineLR(opt, total_steps=10)
    for _ in range(10):
        sched.step()
    assert opt.param_groups[0]["lr"] == pytest.approx(0.0, abs=1e-6)


def test_layernorm_cpu_matches_torch():
    torch.manual_seed(0)
    ln = HipLayerNorm(16)
    x = torch.randn(4, 16)
    out = ln(x)
    ref = torch.nn.functional.layer_norm(x, (16,), ln.weight, ln.bias)
    assert torch.allclose(out, ref, atol=1e-5)


def test_dropout_validation_and_eval_mode():
    with pytest.raises(ValueError):
        HipDropout(1.0)
    d = HipDropout(0.5)
    d.eval()
    x = torch.randn(4, 4)
    assert torch.equal(d(x), x)


def test_hip_conv1x1_cpu_fallback():
    import torch
    from adanet_amd.ops.conv import HipConv1x1
    m = HipConv1x1(8, 12)
    x = torch.randn(2, 8, 5, 5)
    y = m(x)
    ref = torch.nn.functional.conv2d(
        x, m.weight, m.bias.to(x.dtype))
    assert torch.allclose(y, ref, atol=1e-5)
