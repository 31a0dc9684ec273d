"""Refcheck + A/B the LDS-staged depthwise kernels against torch fp32 and
the previous register-tap kernels (toggle ADANET_DW_OLD=1).

The old kernels re-read each input pixel KS^2 times through L1 — fwd<5>
measured 141 us at a shape whose once-through traffic is ~9 us
(profiles/nasprof5_summary.txt). The LDS kernels stage whole (b,c) image
groups once and compute from LDS.

python benchmarks/depthwise_bench.py [--out gpurun_out/dw.json]
"""

import argparse
import json
import sys
import time

import torch
import torch.nn.functional as F

sys.path.insert(0, ".")
from adanet_amd.ops import _extension  # noqa: E402

ext = _extension.require()
dev = "cuda:0"

# (B, C, H, KS, stride) — the NASNet cell shapes (filters 32, cells 3).
SHAPES = [
    (256, 32, 32, 5, 1), (256, 32, 32, 3, 1), (256, 64, 16, 5, 1),
    (256, 64, 16, 3, 1), (256, 160, 8, 5, 1), (256, 32, 32, 5, 2),
    (256, 64, 16, 7, 2), (256, 64, 16, 7, 1), (64, 48, 17, 5, 1),
]


def run_all(B, C, H, KS, stride):
    pad = KS // 2
    torch.manual_seed(B + C + H + KS + stride)
    x = (torch.randn(B, C, H, H, device=dev) / 4).to(torch.bfloat16)
    w = (torch.randn(C, 1, KS, KS, device=dev) / 4).to(torch.bfloat16)
    OH = (H + 2 * pad - KS) // stride + 1
    y = torch.empty(B, C, OH, OH, device=dev, dtype=torch.bfloat16)
    ext.depthwise_fwd(x, w, y, stride, pad)
    dy = (torch.randn_like(y.float()) / 4).to(torch.bfloat16)
    dx = torch.empty_like(x)
    ext.depthwise_bwd_dx(dy, w, dx, stride, pad)
    dw = torch.zeros(C, KS * KS, device=dev)
    ext.depthwise_bwd_dw(x, dy, dw, stride, pad)
    return x, w, y, dy, dx, dw


def refcheck(B, C, H, KS, stride):
    pad = KS // 2
    x, w, y, dy, dx, dw = run_all(B, C, H, KS, stride)
    xf = x.float().requires_grad_(True)
    wf = w.float().requires_grad_(True)
    ref_y = F.conv2d(xf, wf, None, stride, pad, groups=C)
    ref_y.backward(dy.float())
    def rel(a, b):
        return (a.float() - b).abs().max().item() / (
            b.abs().max().item() + 1e-6)
    errs = {"y": rel(y, ref_y.detach()), "dx": rel(dx, xf.grad),
            "dw": rel(dw.reshape(C, 1, KS, KS), wf.grad)}
    ok = all(v < 0.05 for v in errs.values())
    return ok, errs


def bench_op(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default=None)
    ap.add_argument("--skip-refcheck", action="store_true")
    args = ap.parse_args()
    res = {"refcheck": [], "perf": []}
    fails = 0
    if not args.skip_refcheck:
        for shp in SHAPES:
            ok, errs = refcheck(*shp)
            res["refcheck"].append({"shape": shp, "ok": ok,
                                    "errs": {k: round(v, 5)
                                             for k, v in errs.items()}})
            if not ok:
                fails += 1
                print("FAIL", shp, errs)
        # bitwise repeat screen (dW determinism)
        for shp in SHAPES[:3]:
            a = run_all(*shp)[5]
            b = run_all(*shp)[5]
            if not torch.equal(a, b):
                fails += 1
                print("FAIL dw bitrepeat", shp)
        print("refcheck fails:", fails)
        if fails:
            sys.exit(1)

    for (B, C, H, KS, stride) in SHAPES[:6]:
        pad = KS // 2
        x = torch.randn(B, C, H, H, device=dev).to(torch.bfloat16)
        w = torch.randn(C, 1, KS, KS, device=dev).to(torch.bfloat16)
        OH = (H + 2 * pad - KS) // stride + 1
        y = torch.empty(B, C, OH, OH, device=dev, dtype=torch.bfloat16)
        dy = torch.randn(B, C, OH, OH, device=dev).to(torch.bfloat16)
        dx = torch.empty_like(x)
        dw = torch.zeros(C, KS * KS, device=dev)
        row = {"shape": [B, C, H, KS, stride],
               "fwd_us": round(bench_op(
                   lambda: ext.depthwise_fwd(x, w, y, stride, pad)), 1),
               "dx_us": round(bench_op(
                   lambda: ext.depthwise_bwd_dx(dy, w, dx, stride, pad)), 1),
               "dw_us": round(bench_op(
                   lambda: ext.depthwise_bwd_dw(x, dy, dw, stride, pad)), 1)}
        print(json.dumps(row))
        res["perf"].append(row)
    if args.out:
        json.dump(res, open(args.out, "w"), indent=1)


if __name__ == "__main__":
    main()
