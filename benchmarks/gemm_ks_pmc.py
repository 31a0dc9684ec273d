"""PMC probe for the K-slab GEMM at bench + 4k shapes."""
import sys
import torch
sys.path.insert(0, ".")
from adanet_amd.ops import _extension
ext = _extension.require()
dev = "cuda:0"
for (M, N, K, v) in ((2048, 2048, 2048, 51), (4096, 4096, 4096, 50)):
    A = torch.randn(M, K, device=dev).to(torch.bfloat16)
    B = torch.randn(N, K, device=dev).to(torch.bfloat16)
    C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
    for _ in range(3):
        ext.gemm_nt8(A, B, C, None, 0, v)
    torch.cuda.synchronize()
    for _ in range(10):
        ext.gemm_nt8(A, B, C, None, 0, v)
    torch.cuda.synchronize()
print("done")
