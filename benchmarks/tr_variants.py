"""tr-GEMM tile/grouping variant sweep at backward shapes."""
import json
import sys
import time

import torch

sys.path.insert(0, ".")

VARIANTS = {0: "128x128 G0", 1: "128x128 G8", 6: "128x128 G4",
            2: "64x64 G0", 3: "64x64 G8", 4: "256x128 G0", 5: "128x256 G0"}
# (label, M, N, K, ta, tb)
CASES = [("dX hid", 2048, 3072, 2048, 0, 1),
         ("dW hid", 2048, 3072, 2048, 1, 1),
         ("dX sq", 2048, 2048, 2048, 0, 1),
         ("dW sq", 2048, 2048, 2048, 1, 1),
         ("dW big", 4096, 4096, 4096, 1, 1)]


def main():
    from adanet_amd.ops import _extension
    ext = _extension.require()
    dev = "cuda:0"
    torch.manual_seed(0)
    out = {}
    for label, M, N, K, ta, tb in CASES:
        A = (torch.randn(K, M) if ta else torch.randn(M, K)).to(dev).to(
            torch.bfloat16)
        B = (torch.randn(K, N) if tb else torch.randn(N, K)).to(dev).to(
            torch.bfloat16)
        C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        Af = A.float().t() if ta else A.float()
        Bf = B.float() if tb else B.float().t()
        ref = Af @ Bf
        res = {}
        for v, vl in VARIANTS.items():
            try:
                ext.gemm_tr_probe(A, B, C, ta, tb, v)
                torch.cuda.synchronize()
            except RuntimeError as e:
                res[vl] = "ERR " + str(e)[:40]
                continue
            rel = ((C.float() - ref).abs().mean() /
                   (ref.abs().mean() + 1e-3)).item()
            if rel > 0.01:
                res[vl] = "WRONG %.4f" % rel
                continue
            for _ in range(8):
                ext.gemm_tr_probe(A, B, C, ta, tb, v)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(30):
                ext.gemm_tr_probe(A, B, C, ta, tb, v)
            torch.cuda.synchronize()
            sec = (time.perf_counter() - t0) / 30
            res[vl] = round(2.0 * M * N * K / sec / 1e12, 1)
        out["%s %dx%dx%d t%d%d" % (label, M, N, K, ta, tb)] = res
    print(json.dumps(out, indent=1))


if __name__ == "__main__":
    main()
