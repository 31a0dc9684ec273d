"""Kernel microbenchmarks (MI355X): GEMM TFLOPS, mixer/xent/optim GB/s.

Run on a GPU box:  python benchmarks/kernels.py [--quick]
"""

import argparse
import json
import sys
import time

import torch


def bench_kernel(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--quick", action="store_true")
    args = p.parse_args()
    sys.path.insert(0, ".")
    from adanet_amd.ops import _extension
    ext = _extension.require()
    dev = "cuda:0"
    out = {}

    shapes = [(4096, 4096, 4096), (2048, 2048, 3072), (2048, 3072, 2048),
              (4096, 2048, 3072), (8192, 8192, 8192)]
    if args.quick:
        shapes = shapes[:4]
    for (M, N, K) in shapes:
        a = torch.randn(M, K, device=dev).to(torch.bfloat16)
        b = torch.randn(N, K, device=dev).to(torch.bfloat16)
        c = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        sec = bench_kernel(lambda: ext.gemm_nt_bf16(a, b, c, None, 0))
        tf = 2.0 * M * N * K / sec / 1e12
        out["gemm_nt %dx%dx%d" % (M, N, K)] = round(tf, 1)
        # hipBLASLt / rocBLAS comparison via torch.matmul (library GEMM)
        bt = b.t()
        sec2 = bench_kernel(lambda: torch.matmul(a, bt))
        out["torch.matmul %dx%dx%d" % (M, N, K)] = round(
            2.0 * M * N * K / sec2 / 1e12, 1)

    # transpose bandwidth
    x = torch.randn(8192, 8192, device=dev).to(torch.bfloat16)
    y = torch.empty(8192, 8192, device=dev, dtype=torch.bfloat16)
    sec = bench_kernel(lambda: ext.transpose_bf16(x, y))
    out["transpose 8192^2 GB/s"] = round(2 * x.numel() * 2 / sec / 1e9, 1)

    # fused sgd bandwidth (read g+m+p, write m+p+master ~ 20B/elem fp32+bf16)
    n = 64 * 1024 * 1024
    master = torch.randn(n, device=dev, dtype=torch.float32)
    param = master.to(torch.bfloat16)
    grad = torch.randn(n, device=dev).to(torch.bfloat16)
    mom = torch.zeros(n, device=dev, dtype=torch.float32)
    sec = bench_kernel(
        lambda: ext.fused_sgd(master, param, grad, mom, 0.1, 0.9, 0.0, 0.0,
                              False, 1.0), iters=20)
    out["fused_sgd 64M GB/s"] = round(n * 22 / sec / 1e9, 1)

    # xent
    B, C = 16384, 10
    logits = torch.randn(B, C, device=dev).to(torch.bfloat16)
    labels = torch.randint(0, C, (B,), device=dev)
    loss = torch.empty(B, device=dev, dtype=torch.float32)
    probs = torch.empty(B, C, device=dev, dtype=torch.bfloat16)
    sec = bench_kernel(
        lambda: ext.softmax_xent_fwd(logits, labels, loss, probs, 0.0, None))
    out["xent fwd 16k x 10 us"] = round(sec * 1e6, 1)

    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
