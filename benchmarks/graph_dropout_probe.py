"""Minimal repro: HipLinear(+fused dropout) fwd+bwd+FusedSGD under hipGraph
capture vs eager. Prints loss trajectory + param norms."""
import sys
import torch

sys.path.insert(0, ".")
from adanet_amd.ops.linear import HipLinear, restore_fp32_params, \
    direct_grad_writes
from adanet_amd.ops.optim import FusedSGD
from adanet_amd.head import MultiClassHead

dev = torch.device("cuda:0")


def run(graphed, dropout):
    torch.manual_seed(5)
    m = torch.nn.Sequential(
        HipLinear(3072, 2048, activation="relu", dropout=dropout),
        HipLinear(2048, 10)).to(dev).to(torch.bfloat16)
    restore_fp32_params(m)
    head = MultiClassHead(10)
    opt = FusedSGD(m.parameters(), lr=0.05, momentum=0.9)
    opt._overwrite_grads = True
    x = torch.randn(2048, 3072, device=dev).to(torch.bfloat16)
    y = (x.float() @ torch.randn(3072, 10, device=dev)).argmax(1)
    sx = x.clone()
    sy = y.clone()
    losses = []

    def step():
        out = m(sx)
        loss = head.loss(out, sy)
        opt.zero_grad(set_to_none=True)
        with direct_grad_writes():
            loss.backward()
        opt.step()
        return loss.detach()

    if graphed:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                losses.append(step())
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            static_loss = step()
        for i in range(30):
            g.replay()
            if i % 5 == 0:
                losses.append(static_loss.clone())
    else:
        for i in range(32):
            l = step()
            if i % 5 == 0:
                losses.append(l)
    torch.cuda.synchronize()
    vals = [round(float(l), 4) for l in losses]
    wnorm = float(m[0].weight.float().norm())
    print(f"graphed={graphed} dropout={dropout} losses={vals} wnorm={wnorm:.2f}")


run(False, 0.0)
run(False, 0.1)
run(True, 0.0)
run(True, 0.1)
