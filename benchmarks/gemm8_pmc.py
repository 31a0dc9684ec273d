"""PMC probe for the 8-phase GEMM: run each variant a few times at 4096^3
so rocprofv3 --pmc can attribute counters per kernel.

rocprofv3 --pmc SQ_INSTS_MFMA,SQ_INSTS_VALU,SQ_LDS_BANK_CONFLICT,SQ_WAVES \
  -d out -- python benchmarks/gemm8_pmc.py
"""

import sys

import torch

sys.path.insert(0, ".")
from adanet_amd.ops import _extension  # noqa: E402

ext = _extension.require()
dev = "cuda:0"
M = N = K = 4096
A = torch.randn(M, K, device=dev).to(torch.bfloat16)
B = torch.randn(N, K, device=dev).to(torch.bfloat16)
C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)

for _ in range(3):  # warmup, old dispatch only
    ext.gemm_nt_bf16(A, B, C, None, 0)
torch.cuda.synchronize()
for _ in range(10):
    ext.gemm_nt8(A, B, C, None, 0, 0)    # 8ph 256^2 prio
torch.cuda.synchronize()
for _ in range(10):
    ext.gemm_nt8(A, B, C, None, 0, 20)   # 8ph 256^2 noprio
torch.cuda.synchronize()
for _ in range(10):
    ext.gemm_nt_bf16(A, B, C, None, 0)   # old dispatch 256x128
torch.cuda.synchronize()
print("done")
