"""Mid-tile variant sweep at the bench's dominant fwd shapes."""
import json
import sys
import time

import torch

sys.path.insert(0, ".")

VARIANTS = {9: "64x64 mw6 (current)", 47: "64x64 mw4", 48: "64x64 mw8",
            41: "96x96 mw4", 42: "96x96 mw4 G8", 46: "96x96 mw2",
            43: "64x128 4w", 44: "128x64 4w", 45: "64x128 8w",
            20: "128x128 16w G0", 33: "128x128 16w G8"}
SHAPES = [(2048, 2048, 3072), (2048, 3072, 2048), (2048, 2048, 2048)]


def main():
    from adanet_amd.ops import _extension
    ext = _extension.require()
    dev = "cuda:0"
    torch.manual_seed(0)
    results = {}
    for (M, N, K) in SHAPES:
        A = torch.randn(M, K, device=dev).to(torch.bfloat16)
        B = torch.randn(N, K, device=dev).to(torch.bfloat16)
        C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        ref = A.float() @ B.float().t()
        shape_res = {}
        for v, label in VARIANTS.items():
            try:
                ext.gemm_nt_bf16_probe(A, B, C, v)
                torch.cuda.synchronize()
            except RuntimeError as e:
                shape_res[label] = "ERR %s" % str(e)[:30]
                continue
            rel = ((C.float() - ref).abs().mean() /
                   (ref.abs().mean() + 1e-3)).item()
            if rel > 0.01:
                shape_res[label] = "WRONG %.4f" % rel
                continue
            for _ in range(8):
                ext.gemm_nt_bf16_probe(A, B, C, v)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(30):
                ext.gemm_nt_bf16_probe(A, B, C, v)
            torch.cuda.synchronize()
            sec = (time.perf_counter() - t0) / 30
            shape_res[label] = round(2.0 * M * N * K / sec / 1e12, 1)
        results["%dx%dx%d" % (M, N, K)] = shape_res
    print(json.dumps(results, indent=1))


if __name__ == "__main__":
    main()
