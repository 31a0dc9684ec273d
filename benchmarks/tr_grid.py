"""A/B: transposed-staging GEMM vs transpose+NT composite at backward shapes."""
import json
import sys
import time

import torch

sys.path.insert(0, ".")


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
    from adanet_amd.ops import _extension
    ext = _extension.require()
    dev = "cuda:0"
    torch.manual_seed(0)
    # (label, M, N, K, trans_a, trans_b)
    cases = [
        ("dX hid 2048x3072 k2048", 2048, 3072, 2048, 0, 1),
        ("dW hid 2048x3072 k2048(batch)", 2048, 3072, 2048, 1, 1),
        ("dW wide 4096-unit", 4096, 2048, 2048, 1, 1),
        ("dX wide", 2048, 2048, 4096, 0, 1),
        ("big 4096x4096 k4096 trb", 4096, 4096, 4096, 0, 1),
        ("big 4096x4096 k4096 tt", 4096, 4096, 4096, 1, 1),
    ]
    out = {}
    for label, M, N, K, ta, tb in cases:
        A = (torch.randn(K, M) if ta else torch.randn(M, K)).to(dev).to(torch.bfloat16)
        B = (torch.randn(K, N) if tb else torch.randn(N, K)).to(dev).to(torch.bfloat16)
        C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        ext.gemm_tr_bf16(A, B, C, None, 0, ta, tb)
        torch.cuda.synchronize()
        Af = A.float().t() if ta else A.float()
        Bf = B.float() if tb else B.float().t()
        ref = Af @ Bf
        rel = ((C.float() - ref).abs().mean() / (ref.abs().mean() + 1e-3)).item()
        if rel > 0.01:
            out[label] = "WRONG %.4f" % rel
            continue
        sec = bench(lambda: ext.gemm_tr_bf16(A, B, C, None, 0, ta, tb))
        tf_tr = 2.0 * M * N * K / sec / 1e12

        # composite: transposes + NT
        def composite():
            An = A
            Bn = B
            if ta:
                An = torch.empty(M, K, device=dev, dtype=torch.bfloat16)
                ext.transpose_bf16(A, An)
            if tb:
                Bn = torch.empty(N, K, device=dev, dtype=torch.bfloat16)
                ext.transpose_bf16(B, Bn)
            ext.gemm_nt_bf16(An, Bn, C, None, 0)
        sec2 = bench(composite)
        tf_comp = 2.0 * M * N * K / sec2 / 1e12
        out[label] = {"tr_TF": round(tf_tr, 1),
                      "composite_TF": round(tf_comp, 1),
                      "speedup": round(tf_tr / tf_comp, 3)}
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
