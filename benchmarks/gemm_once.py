"""Launch the dispatch GEMM a few times for PMC counter capture."""
import sys
import torch
sys.path.insert(0, ".")
from adanet_amd.ops import _extension
ext = _extension.require()
dev = "cuda:0"
for (M, N, K) in [(4096, 4096, 4096), (2048, 2048, 3072)]:
    a = torch.randn(M, K, device=dev).to(torch.bfloat16)
    b = torch.randn(N, K, device=dev).to(torch.bfloat16)
    c = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
    for _ in range(10):
        ext.gemm_nt_bf16(a, b, c, None, 0)
    torch.cuda.synchronize()
print("done")
