"""A/B the GEMM tile/occupancy/K-step variants in one GPU session.

python benchmarks/gemm_variants.py
Prints TFLOPS per (variant, shape) with correctness check vs fp32 torch.
"""

import json
import sys
import time

import torch

VARIANTS = {
    0: "128x128 f4x4 mw2 k32 (base)",
    1: "128x128 f4x4 mw3 k32",
    2: "128x128 f4x4 mw4 k32",
    3: "64x64 f2x2 mw4 k32 (base small)",
    4: "64x64 f2x2 mw8 k32",
    5: "128x64 f4x2 mw4 k32",
    6: "128x64 f4x2 mw2 k32",
    7: "128x128 f4x4 mw2 k64",
    8: "128x64 f4x2 mw4 k64",
    9: "64x64 f2x2 mw6 k32",
    10: "64x128 f2x4 mw4 k32",
    11: "128x128 mw2 splitk2",
    12: "128x128 mw2 splitk4",
    13: "128x128 mw4 splitk4",
    14: "128x128 mw4 splitk2",
    15: "64x64 mw6 splitk2",
    16: "64x64 f2x2 mw6 k64",
    17: "64x64 f2x2 mw4 k64",
    18: "64x128 f2x4 mw4 k64",
    19: "64x128 f2x4 mw2 k64",
    20: "128x128 16w (4x4 grid, 32x32/wave) mw2",
    21: "128x128 8w (2x4 grid, 64x32/wave) mw2",
    22: "128x128 8w (4x2 grid, 32x64/wave) mw2",
    23: "256x128 8w (4x2 grid, 64x64/wave) mw2",
    24: "128x256 8w (2x4 grid, 64x64/wave) mw2",
    25: "128x128 16w mw4",
    26: "256x128 16w (64x32/wave) mw2",
    27: "256x256 16w (64x64/wave) mw2",
    28: "256x64 8w (64x32/wave) mw2",
    29: "256x128 8w k64",
    30: "128x128 16w k64",
}

SHAPES = [(2048, 2048, 3072), (4096, 2048, 3072), (2048, 3072, 2048),
          (4096, 4096, 4096)]


def bench(fn, iters=30, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    sys.path.insert(0, ".")
    from adanet_amd.ops import _extension
    ext = _extension.require()
    dev = "cuda:0"
    out = {}
    for (M, N, K) in SHAPES:
        torch.manual_seed(M + N)
        a = torch.randn(M, K, device=dev).to(torch.bfloat16)
        b = torch.randn(N, K, device=dev).to(torch.bfloat16)
        c = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        ref = (a.float() @ b.float().t())
        refscale = ref.abs().mean().item() + 1e-3
        for v, name in VARIANTS.items():
            try:
                ext.gemm_nt_bf16_probe(a, b, c, v)
                torch.cuda.synchronize()
                err = ((c.float() - ref).abs().mean() / refscale).item()
                if err > 0.01:
                    out["%dx%dx%d v%d" % (M, N, K, v)] = "WRONG %.4f" % err
                    continue
                sec = bench(lambda: ext.gemm_nt_bf16_probe(a, b, c, v))
                tf = 2.0 * M * N * K / sec / 1e12
                out["%dx%dx%d v%-2d %s" % (M, N, K, v, name)] = round(tf, 1)
            except Exception as e:
                out["%dx%dx%d v%d" % (M, N, K, v)] = "ERR %s" % str(e)[:60]
        bt = b.t()
        sec = bench(lambda: torch.matmul(a, bt))
        out["%dx%dx%d hipblaslt" % (M, N, K)] = round(
            2.0 * M * N * K / sec / 1e12, 1)
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
