import sys, torch
sys.path.insert(0, ".")
from adanet_amd.ops import _extension
ext = _extension.require()
dev = "cuda:0"
fails = 0
for (M, N, K) in ((32, 288, 16384), (64, 64, 8192), (160, 1440, 65536),
                  (48, 300, 4096)):
    torch.manual_seed(M + K)
    A = (torch.randn(M, K, device=dev) / 8).to(torch.bfloat16)
    B = (torch.randn(N, K, device=dev) / 8).to(torch.bfloat16)
    C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
    outs = []
    for rep in range(4):
        C.zero_()
        ext.gemm_nt_bf16(A, B, C, None, 0)
        outs.append(C.clone())
    ref = A.float() @ B.float().t()
    rel = (outs[0].float() - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    bit = all(torch.equal(outs[0], o) for o in outs[1:])
    ok = rel < 0.05 and bit
    print(M, N, K, "rel", round(rel, 5), "bit", bit, "OK" if ok else "FAIL")
    fails += 0 if ok else 1
sys.exit(1 if fails else 0)
