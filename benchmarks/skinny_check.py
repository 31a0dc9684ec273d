import sys, time, torch
sys.path.insert(0, ".")
from adanet_amd.ops import _extension
ext = _extension.require()
dev = "cuda:0"
fails = 0
for (M, N, K, act) in ((2048, 10, 2048, 0), (2048, 10, 3072, 1),
                       (2048, 16, 2048, 0), (300, 10, 2048, 0),
                       (2048, 10, 2048, 2)):
    torch.manual_seed(M + N + K + act)
    A = (torch.randn(M, K, device=dev) / 8).to(torch.bfloat16)
    B = (torch.randn(N, K, device=dev) / 8).to(torch.bfloat16)
    bias = torch.randn(N, device=dev)
    C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
    C0 = (torch.randn(M, N, device=dev) / 8).to(torch.bfloat16)
    if act == 2:
        C.copy_(C0)
    outs = []
    for rep in range(4):
        if act == 2:
            C.copy_(C0)
        ext.gemm_nt_bf16(A, B, C, bias, act)
        outs.append(C.clone())
    ref = A.float() @ B.float().t() + bias
    if act == 1:
        ref = ref.relu()
    if act == 2:
        ref += C0.float()
    rel = (outs[0].float() - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    bit = all(torch.equal(outs[0], o) for o in outs[1:])
    ok = rel < 0.05 and bit
    print("fwd", M, N, K, "act", act, "rel", round(rel, 5), "bit", bit, "OK" if ok else "FAIL")
    fails += 0 if ok else 1
# dW tt skinny-M via gemm_tr_bf16
# M padded to 16 as HipLinear does for the n_classes dim (strides %8).
for (M, N, K) in ((16, 2048, 2048), (32, 2048, 2048)):
    torch.manual_seed(M + N + K)
    A = (torch.randn(K, M, device=dev) / 8).to(torch.bfloat16)  # ta
    B = (torch.randn(N, K, device=dev) / 8).to(torch.bfloat16)  # tb
    C = torch.empty(M, N, device=dev, dtype=torch.bfloat16)
    ext.gemm_tr_bf16(A, B, C, None, 0, 1, 1)
    ref = A.float().t() @ B.float().t().t().t()
    ref = A.float().t() @ B.float()
    rel = (C.float() - ref).abs().max().item() / (ref.abs().max().item() + 1e-6)
    ok = rel < 0.05
    print("tt", M, N, K, "rel", round(rel, 5), "OK" if ok else "FAIL")
    fails += 0 if ok else 1
# perf
def t(fn, it=50):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(it):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / it * 1e6
A = torch.randn(2048, 2048, device=dev).to(torch.bfloat16)
B = torch.randn(10, 2048, device=dev).to(torch.bfloat16)
C = torch.empty(2048, 10, device=dev, dtype=torch.bfloat16)
print("logits fwd us:", round(t(lambda: ext.gemm_nt_bf16(A, B, C, None, 0)), 2),
      "torch:", round(t(lambda: torch.mm(A, B.t(), out=C)), 2))
At = torch.randn(2048, 16, device=dev).to(torch.bfloat16)
Bt = torch.randn(2048, 2048, device=dev).to(torch.bfloat16)
Ct = torch.empty(16, 2048, device=dev, dtype=torch.bfloat16)
print("logits dW us:", round(t(lambda: ext.gemm_tr_bf16(At, Bt, Ct, None, 0, 1, 1)), 2),
      "torch:", round(t(lambda: torch.mm(At.t(), Bt, out=Ct)), 2))
sys.exit(1 if fails else 0)
