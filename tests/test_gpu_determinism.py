"""Bit-repeatability regression tests for the deterministic reductions
(run-to-run reproducibility is a round-2 guarantee: BENCHMARKS.md
'Run-to-run determinism'). Each op runs repeatedly on identical inputs;
outputs must be byte-identical."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _ext():
    from adanet_amd.ops import _extension
    return _extension.require()


@pytest.mark.parametrize("shape", [(2048, 2048), (2048, 2049), (64, 10)])
def test_relu_bwd_colsum_bitrepeat(shape):
    ext = _ext()
    B, C = shape
    torch.manual_seed(3)
    dy = (torch.randn(B, C, device="cuda") / 4).to(torch.bfloat16)
    y = torch.randn(B, C, device="cuda").to(torch.bfloat16)
    db0 = torch.randn(C, device="cuda")
    ref = None
    for _ in range(5):
        dz = torch.empty_like(dy)
        db = db0.clone()
        ext.relu_bwd_colsum(dy, y, dz, db, 1.0)
        if ref is None:
            ref = (dz.clone(), db.clone())
        else:
            assert torch.equal(ref[0], dz) and torch.equal(ref[1], db)


def test_depthwise_dw_bitrepeat():
    ext = _ext()
    torch.manual_seed(4)
    x = (torch.randn(64, 32, 32, 32, device="cuda") / 4).to(torch.bfloat16)
    dy = (torch.randn(64, 32, 32, 32, device="cuda") / 4).to(torch.bfloat16)
    ref = None
    for _ in range(5):
        dw = torch.empty(32, 25, device="cuda")
        ext.depthwise_bwd_dw(x, dy, dw, 1, 2)
        if ref is None:
            ref = dw.clone()
        else:
            assert torch.equal(ref, dw)


def test_splitk_bitrepeat_and_no_empty_slots():
    # ksplit > busy-splits regression: M=32, K=16384 drives ksplit past
    # the K-tile count; unwritten C32 slots made dW nondeterministic.
    ext = _ext()
    torch.manual_seed(5)
    A = (torch.randn(32, 16384, device="cuda") / 8).to(torch.bfloat16)
    B = (torch.randn(288, 16384, device="cuda") / 8).to(torch.bfloat16)
    ref = None
    for _ in range(5):
        C = torch.zeros(32, 288, device="cuda", dtype=torch.bfloat16)
        ext.gemm_nt_bf16(A, B, C, None, 0)
        if ref is None:
            ref = C.clone()
        else:
            assert torch.equal(ref, C)
    rel = (ref.float() - A.float() @ B.float().t()).abs().max().item()
    assert rel < 0.05 * (A.float() @ B.float().t()).abs().max().item()


def test_batchnorm_bwd_bitrepeat():
    ext = _ext()
    torch.manual_seed(6)
    x = (torch.randn(64, 32, 16, 16, device="cuda") / 4).to(torch.bfloat16)
    dy = (torch.randn(64, 32, 16, 16, device="cuda") / 4).to(torch.bfloat16)
    mean = torch.randn(32, device="cuda") / 10
    rstd = torch.rand(32, device="cuda") + 0.5
    ref = None
    for _ in range(5):
        dx = torch.empty_like(x)
        sdy = torch.empty(32, device="cuda")
        sdyx = torch.empty(32, device="cuda")
        ext.batchnorm_bwd(x, dy, dx, mean, rstd, None, sdy, sdyx)
        if ref is None:
            ref = (dx.clone(), sdy.clone(), sdyx.clone())
        else:
            assert torch.equal(ref[0], dx)
            assert torch.equal(ref[1], sdy)
            assert torch.equal(ref[2], sdyx)
