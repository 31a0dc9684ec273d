"""Refcheck + A/B the 8-phase GEMM (gemm_nt8) against the round-1 kernel
and hipBLASLt (torch.mm) in one GPU session.

python benchmarks/gemm8_bench.py [--out gpurun_out/gemm8.json]

Stage 1 (refcheck): every variant (incl. SAFE drain twins) vs fp32 torch
at small sizes + edge tiles. Any mismatch prints FAIL and exits nonzero —
perf numbers from a racing kernel are meaningless.
Stage 2 (perf): interleaved A/B across shapes: old dispatch, 8-phase
variants, torch.mm (hipBLASLt).
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


def refcheck(M, N, K, variant, act=0, bias=False, atol=0.1):
    torch.manual_seed(42 + M + N + K + variant)
    A = (torch.randn(M, K, device=dev) / 8).to(torch.bfloat16)
    B = (torch.randn(N, K, device=dev) / 8).to(torch.bfloat16)
    bias_t = torch.randn(N, device=dev) if bias else None
    C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
    if act == 2:
        C0 = (torch.randn(M, N, device=dev) / 8).to(torch.bfloat16)
        C.copy_(C0)
    ext.gemm_nt8(A, B, C, bias_t, act, variant)
    ref = A.float() @ B.float().t()
    if bias:
        ref += bias_t
    if act == 1:
        ref = ref.relu()
    if act == 2:
        ref += C0.float()
    err = (C.float() - ref).abs().max().item()
    scale = ref.abs().max().item() + 1e-6
    ok = err / scale < atol
    return ok, err / scale


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
        cases = []
        for variant in (0, 3, 20, 30, 50, 51, 52, 53):
            for (M, N, K) in ((256, 256, 64), (512, 512, 128),
                              (1024, 768, 256), (512, 512, 64)):
                cases.append((M, N, K, variant, 0, False))
            # edge tiles (M/N not multiples of the tile)
            cases.append((300, 200, 128, variant, 0, False))
            cases.append((513, 257, 192, variant, 0, True))
            # epilogues
            cases.append((512, 512, 128, variant, 1, True))
            cases.append((512, 512, 128, variant, 2, False))
        for (M, N, K, v, act, bias) in cases:
            ok, rel = refcheck(M, N, K, v, act, bias)
            results["refcheck"].append(
                {"MNK": [M, N, K], "variant": v, "act": act, "bias": bias,
                 "ok": ok, "rel_err": round(rel, 5)})
            if not ok:
                fails += 1
                print("FAIL", M, N, K, "variant", v, "act", act, "rel", rel)
        # multi-run race screen on the steady-state variant (guide m152):
        for v in (0, 50, 51):
            for rep in range(8):
                ok, rel = refcheck(4096, 4096, 4096, v, 0, False)
                results["refcheck"].append(
                    {"MNK": [4096, 4096, 4096], "variant": v, "rep": rep,
                     "ok": ok, "rel_err": round(rel, 5)})
                if not ok:
                    fails += 1
                    print("FAIL 4096 v", v, "rep", rep, "rel", rel)
        print("refcheck fails:", fails)
        if fails:
            if args.out:
                json.dump(results, open(args.out, "w"), indent=1)
            sys.exit(1)

    shapes = [(2048, 2048, 2048), (2048, 2048, 3072), (4096, 4096, 4096),
              (8192, 8192, 8192)]
    for (M, N, K) in shapes:
        A = torch.randn(M, K, device=dev).to(torch.bfloat16)
        B = torch.randn(N, K, device=dev).to(torch.bfloat16)
        C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        Bt = B.t().contiguous().t()  # K-major view for torch.mm parity
        flops = 2.0 * M * N * K
        row = {"MNK": [M, N, K]}

        def tf(dt):
            return round(flops / dt / 1e12, 1)

        row["old_dispatch"] = tf(bench_fn(
            lambda: ext.gemm_nt_bf16(A, B, C, None, 0), args.iters))
        for v, name in ((3, "8ph_128x128"), (20, "8ph_256_noprio"),
                        (50, "ks_256x256"), (51, "ks_128x128_8w"),
                        (52, "ks_128x128_4w"), (53, "ks_256x128")):
            try:
                row[name] = tf(bench_fn(
                    lambda v=v: ext.gemm_nt8(A, B, C, None, 0, v),
                    args.iters))
            except Exception as e:
                row[name] = str(e)[:60]
        row["torch_mm"] = tf(bench_fn(
            lambda: torch.mm(A, B.t(), out=C), args.iters))
        print(json.dumps(row))
        results["perf"].append(row)
        del A, B, C, Bt
        torch.cuda.empty_cache()

    if args.out:
        json.dump(results, open(args.out, "w"), indent=1)


if __name__ == "__main__":
    main()
