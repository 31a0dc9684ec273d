"""Refcheck the act=3 (fused relu+dropout) epilogue on EVERY dispatch path:
gemv (M<=8), old tiled, nt8 128^2, nt8 256^2, generic. Verifies each output
is either 0 or relu_ref/(1-p), kept fraction ~= 1-p, and determinism for a
fixed seed."""
import sys

import torch

sys.path.insert(0, ".")
from adanet_amd.ops import _extension

ext = _extension.require()
dev = "cuda:0"
p = 0.3
seed = torch.tensor([12345], dtype=torch.int64, device=dev)

shapes = [
    ("gemv", 4, 512, 256),
    ("old_tile", 512, 512, 256),
    ("nt8_128", 2048, 2048, 2048),
    ("nt8_256", 4096, 4096, 512),
    ("generic", 33, 70, 40),
]
fails = 0
for tag, M, N, K in shapes:
    torch.manual_seed(M + N)
    A = (torch.randn(M, K, device=dev) / 8).to(torch.bfloat16)
    B = (torch.randn(N, K, device=dev) / 8).to(torch.bfloat16)
    C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
    ext.gemm_nt_bf16(A, B, C, None, 3, p, seed)
    ref = torch.relu(A.float() @ B.float().t())
    c = C.float()
    kept = c != 0
    pos = ref > 1e-2
    scaled = ref / (1.0 - p)
    ok_vals = ((c - scaled).abs() < 0.05 * (1 + scaled.abs())) | (~kept)
    frac = kept[pos].float().mean().item() if pos.any() else 1.0
    C2 = torch.empty_like(C)
    ext.gemm_nt_bf16(A, B, C2, None, 3, p, seed)
    det = torch.equal(C, C2)
    bad = (~ok_vals).sum().item()
    ok = bad == 0 and abs(frac - (1 - p)) < 0.05 and det
    print(tag, "ok" if ok else "FAIL", "badvals", bad,
          "kept_frac", round(frac, 3), "det", det,
          "nan", torch.isnan(c).any().item())
    fails += 0 if ok else 1
print("fails:", fails)
sys.exit(1 if fails else 0)
