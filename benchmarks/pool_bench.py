"""Refcheck + time the stride-templated pool3 kernels vs torch fp32.

python benchmarks/pool_bench.py [--out gpurun_out/pool.json]
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

SHAPES = [(256, 32, 32, 1), (256, 64, 16, 1), (256, 160, 8, 1),
          (256, 32, 32, 2), (256, 64, 16, 2), (64, 48, 17, 1)]


def refcheck(B, C, H, stride, is_max):
    torch.manual_seed(B + C + H + stride + is_max)
    x = (torch.randn(B, C, H, H, device=dev) / 4).to(torch.bfloat16)
    OH = (H + 2 - 3) // stride + 1
    y = torch.empty(B, C, OH, OH, device=dev, dtype=torch.bfloat16)
    am = torch.empty(B, C, OH, OH, device=dev, dtype=torch.uint8) \
        if is_max else None
    ext.pool3_fwd(x, y, am, stride, is_max)
    dy = (torch.randn_like(y.float()) / 4).to(torch.bfloat16)
    dx = torch.empty_like(x)
    ext.pool3_bwd(dy, am, dx, stride, is_max)
    xf = x.float().requires_grad_(True)
    if is_max:
        ref = F.max_pool2d(xf, 3, stride, 1)
    else:
        ref = F.avg_pool2d(xf, 3, stride, 1, count_include_pad=False)
    ref.backward(dy.float())
    ry = (y.float() - ref.detach()).abs().max().item()
    rdx = (dx.float() - xf.grad).abs().max().item()
    scale = ref.detach().abs().max().item() + 1e-6
    gs = xf.grad.abs().max().item() + 1e-6
    return ry / scale < 0.05 and rdx / gs < 0.05, (ry / scale, rdx / gs)


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
    args = ap.parse_args()
    res = {"refcheck": [], "perf": []}
    fails = 0
    for shp in SHAPES:
        for is_max in (0, 1):
            ok, errs = refcheck(*shp, is_max)
            res["refcheck"].append({"shape": shp, "max": is_max, "ok": ok})
            if not ok:
                fails += 1
                print("FAIL", shp, is_max, errs)
    print("refcheck fails:", fails)
    if fails:
        sys.exit(1)
    for (B, C, H, stride) in SHAPES[:5]:
        x = torch.randn(B, C, H, H, device=dev).to(torch.bfloat16)
        OH = (H + 2 - 3) // stride + 1
        y = torch.empty(B, C, OH, OH, device=dev, dtype=torch.bfloat16)
        am = torch.empty(B, C, OH, OH, device=dev, dtype=torch.uint8)
        dy = torch.randn(B, C, OH, OH, device=dev).to(torch.bfloat16)
        dx = torch.empty_like(x)
        row = {"shape": [B, C, H, stride],
               "fwd_avg_us": round(bench_op(
                   lambda: ext.pool3_fwd(x, y, None, stride, 0)), 1),
               "fwd_max_us": round(bench_op(
                   lambda: ext.pool3_fwd(x, y, am, stride, 1)), 1),
               "bwd_avg_us": round(bench_op(
                   lambda: ext.pool3_bwd(dy, None, dx, stride, 0)), 1),
               "bwd_max_us": round(bench_op(
                   lambda: ext.pool3_bwd(dy, am, dx, stride, 1)), 1)}
        print(json.dumps(row))
        res["perf"].append(row)
    if args.out:
        json.dump(res, open(args.out, "w"), indent=1)


if __name__ == "__main__":
    main()
