"""L2 supertile-grouping (GROUPM) A/B at bench shapes."""
import json
import sys
import time

import torch

sys.path.insert(0, ".")

# variant -> label
VARIANTS = {
    20: "128x128/16w G0",
    31: "128x128/16w G2",
    32: "128x128/16w G4",
    33: "128x128/16w G8",
    34: "128x128/16w G16",
    23: "256x128/8w G0",
    35: "256x128/8w G2",
    36: "256x128/8w G4",
    9:  "64x64/6mw G0",
    37: "64x64 G4",
    38: "64x64 G8",
    24: "128x256/8w G0",
    39: "128x256 G2",
    40: "128x256 G4",
}

SHAPES = [(2048, 2048, 3072), (2048, 3072, 2048), (2048, 2048, 2048),
          (4096, 4096, 4096)]


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
        ref = None
        shape_res = {}
        for v, label in VARIANTS.items():
            try:
                ext.gemm_nt_bf16_probe(A, B, C, v)
                torch.cuda.synchronize()
            except RuntimeError as e:
                shape_res[label] = "ERR %s" % str(e)[:40]
                continue
            if ref is None:
                ref = A.float() @ B.float().t()
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
