"""Refcheck + A/B the generalized pipelined GEMM (gemm_x8: K-major operand
staging) against the round-1 tr kernels on the backward dX/dW shapes.

python benchmarks/gemm_x8_bench.py [--out gpurun_out/gemm_x8.json]
"""

import argparse
import json
import sys
import time

import torch

sys.path.insert(0, ".")
from adanet_amd.ops import _extension  # noqa: E402

ext = _extension.require()
dev = "cuda:0"


def make(M, N, K, ta, tb, seed):
    torch.manual_seed(seed)
    Af = torch.randn(M, K, device=dev) / 8
    Bf = torch.randn(N, K, device=dev) / 8
    A = (Af.t().contiguous() if ta else Af).to(torch.bfloat16)
    B = (Bf.t().contiguous() if tb else Bf).to(torch.bfloat16)
    return A, B, Af, Bf


def refcheck(M, N, K, ta, tb, variant, act=0, bias=False, atol=0.1):
    A, B, Af, Bf = make(M, N, K, ta, tb, 91 + M + N + K + variant)
    bias_t = torch.randn(N, device=dev) if bias else None
    C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
    if act == 2:
        C0 = (torch.randn(M, N, device=dev) / 8).to(torch.bfloat16)
        C.copy_(C0)
    ext.gemm_x8(A, B, C, bias_t, act, int(ta), int(tb), variant)
    a2 = A.float().t() if ta else A.float()
    b2 = B.float() if tb else B.float().t()
    ref = a2 @ b2
    if bias:
        ref += bias_t
    if act == 1:
        ref = ref.relu()
    if act == 2:
        ref += C0.float()
    err = (C.float() - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-6
    return err / scale < atol, err / scale


def bench_fn(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default=None)
    ap.add_argument("--skip-refcheck", action="store_true")
    ap.add_argument("--iters", type=int, default=20)
    args = ap.parse_args()
    results = {"refcheck": [], "perf": []}

    if not args.skip_refcheck:
        fails = 0
        for variant in (0, 1, 10, 11, 12):
            for (ta, tb) in ((True, True), (False, True), (True, False),
                             (False, False)):
                for (M, N, K) in ((256, 256, 128), (512, 384, 192),
                                  (304, 200, 128)):
                    ok, rel = refcheck(M, N, K, ta, tb, variant)
                    results["refcheck"].append(
                        {"MNK": [M, N, K], "ta": ta, "tb": tb, "v": variant,
                         "ok": ok, "rel": round(rel, 5)})
                    if not ok:
                        fails += 1
                        print("FAIL", M, N, K, ta, tb, "v", variant, rel)
            # epilogues
            ok, rel = refcheck(512, 512, 128, True, True, variant, 2, False)
            if not ok:
                fails += 1
                print("FAIL act2 v", variant, rel)
            ok, rel = refcheck(512, 512, 128, False, True, variant, 1, True)
            if not ok:
                fails += 1
                print("FAIL act1 v", variant, rel)
        # race screen on the dW shape
        for v in (0, 10):
            for rep in range(6):
                ok, rel = refcheck(2048, 2048, 2048, True, True, v)
                if not ok:
                    fails += 1
                    print("FAIL race tt v", v, rep, rel)
        print("refcheck fails:", fails)
        if fails:
            if args.out:
                json.dump(results, open(args.out, "w"), indent=1)
            sys.exit(1)

    # perf on the bench's backward shapes
    shapes = [
        ("dW", 2048, 2048, 2048, True, True),
        ("dW_in", 3072, 2048, 2048, True, True),
        ("dX", 2048, 3072, 2048, False, True),
        ("dX_h", 2048, 2048, 2048, False, True),
        ("tt4k", 4096, 4096, 4096, True, True),
    ]
    for (tag, M, N, K, ta, tb) in shapes:
        A, B, _, _ = make(M, N, K, ta, tb, 7)
        C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        flops = 2.0 * M * N * K
        row = {"tag": tag, "MNK": [M, N, K], "ta": ta, "tb": tb}

        def tf(dt):
            return round(flops / dt / 1e12, 1)

        row["old_tr"] = tf(bench_fn(
            lambda: ext.gemm_tr_bf16(A, B, C, None, 0, int(ta), int(tb)),
            args.iters))
        for v, name in ((0, "x8_128_8w"), (10, "ks_128_8w"),
                        (11, "ks_256"), (12, "ks_256x128")):
            try:
                row[name] = tf(bench_fn(
                    lambda v=v: ext.gemm_x8(A, B, C, None, 0, int(ta),
                                            int(tb), v), args.iters))
            except Exception as e:
                row[name] = str(e)[:50]
        # hipBLASLt comparison (it transposes via strides, no copy)
        a2 = A.t() if ta else A
        b2 = B if tb else B.t()
        row["torch_mm"] = tf(bench_fn(
            lambda: torch.mm(a2, b2, out=C), args.iters))
        print(json.dumps(row))
        results["perf"].append(row)
        del A, B, C
        torch.cuda.empty_cache()

    if args.out:
        json.dump(results, open(args.out, "w"), indent=1)


if __name__ == "__main__":
    main()
