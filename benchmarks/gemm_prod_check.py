"""Production-dispatch GEMM timing at the bench's hot shapes."""
import sys
import time

import torch

sys.path.insert(0, ".")


def main():
    from adanet_amd.ops import _extension
    ext = _extension.require()
    dev = "cuda:0"
    torch.manual_seed(0)
    for (tag, M, N, K) in [("fwd1", 2048, 2048, 3072),
                           ("fwd2", 2048, 2048, 2048),
                           ("fwdL1w", 2048, 3072, 2048)]:
        A = torch.randn(M, K, device=dev).to(torch.bfloat16)
        B = torch.randn(N, K, device=dev).to(torch.bfloat16)
        C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        ext.gemm_nt_bf16(A, B, C, None, 0)
        ref = A.float() @ B.float().t()
        rel = ((C.float() - ref).abs().mean() / (ref.abs().mean() + 1e-3)).item()
        for _ in range(8):
            ext.gemm_nt_bf16(A, B, C, None, 0)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(30):
            ext.gemm_nt_bf16(A, B, C, None, 0)
        torch.cuda.synchronize()
        sec = (time.perf_counter() - t0) / 30
        print(tag, M, N, K, "TF", round(2.0*M*N*K/sec/1e12, 1), "rel", round(rel, 5))
    # tr production at dX sq / dW sq
    for (tag, M, N, K, ta, tb) in [("dXsq", 2048, 2048, 2048, 0, 1),
                                   ("dWsq", 2048, 2048, 2048, 1, 1),
                                   ("dXwide", 2048, 4096, 4096, 0, 1)]:
        A = (torch.randn(K, M) if ta else torch.randn(M, K)).to(dev).to(torch.bfloat16)
        B = (torch.randn(K, N) if tb else torch.randn(N, K)).to(dev).to(torch.bfloat16)
        C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        ext.gemm_tr_bf16(A, B, C, None, 0, ta, tb)
        torch.cuda.synchronize()
        for _ in range(8):
            ext.gemm_tr_bf16(A, B, C, None, 0, ta, tb)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(30):
            ext.gemm_tr_bf16(A, B, C, None, 0, ta, tb)
        torch.cuda.synchronize()
        sec = (time.perf_counter() - t0) / 30
        print(tag, "TF", round(2.0*M*N*K/sec/1e12, 1))


if __name__ == "__main__":
    main()
