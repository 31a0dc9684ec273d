"""Probe ds_read_b64_tr_b16 layout + transposed-staging GEMM numerics/perf."""

import json
import sys
import time

import torch

sys.path.insert(0, ".")


def main():
    from adanet_amd.ops import _extension
    ext = _extension.require()
    dev = "cuda:0"

    # 1) layout probe: which ramp value does lane l elem j receive?
    out = torch.zeros(64 * 8, device=dev, dtype=torch.float32)
    ext.probe_tr16_layout(out)
    torch.cuda.synchronize()
    vals = out.cpu().int().reshape(64, 8).tolist()
    print("lane0:", vals[0])
    print("lane1:", vals[1])
    print("lane2:", vals[2])
    print("lane15:", vals[15])
    print("lane16:", vals[16])
    print("lane17:", vals[17])
    print("lane31:", vals[31])
    print("lane32:", vals[32])
    print("lane48:", vals[48])
    # Expected under the quad-transpose model with tr_frag_addrs: lane l
    # elem j == T[8g + j][l&15] == (8*(l>>4)+j)*16 + (l&15).
    want_ok = all(
        vals[l][j] == ((8 * (l >> 4) + j) * 16 + (l & 15)) % 256
        and vals[l][4 + j] == ((8 * (l >> 4) + 4 + j) * 16 + (l & 15)) % 256
        for l in range(64) for j in range(4))
    print("matches fragment mapping T[8g+j][l&15]:", want_ok)

    # 2) numerics of the transposed GEMM probes
    torch.manual_seed(0)
    for (M, N, K, ta, tb) in [(256, 256, 128, 0, 1), (256, 256, 128, 1, 1),
                              (2048, 3072, 2048, 0, 1),
                              (2048, 3072, 2048, 1, 1)]:
        A = (torch.randn(K, M) if ta else torch.randn(M, K)).to(dev).to(
            torch.bfloat16)
        B = (torch.randn(K, N) if tb else torch.randn(N, K)).to(dev).to(
            torch.bfloat16)
        C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
        ext.gemm_tr_probe(A, B, C, ta, tb, 0)
        torch.cuda.synchronize()
        Af = A.float().t() if ta else A.float()
        Bf = B.float() if tb else B.float().t()
        ref = Af @ Bf
        rel = ((C.float() - ref).abs().mean() / (ref.abs().mean() + 1e-3)).item()
        line = {"shape": (M, N, K, ta, tb), "rel_err": round(rel, 5)}
        if rel < 0.01 and M >= 2048:
            for _ in range(5):
                ext.gemm_tr_probe(A, B, C, ta, tb, 0)
            torch.cuda.synchronize()
            t0 = time.perf_counter()
            for _ in range(30):
                ext.gemm_tr_probe(A, B, C, ta, tb, 0)
            torch.cuda.synchronize()
            sec = (time.perf_counter() - t0) / 30
            line["TF"] = round(2.0 * M * N * K / sec / 1e12, 1)
        print(json.dumps(line))


if __name__ == "__main__":
    main()
