"""Per-op bit-repeatability screen for the conv family on NASNet shapes
(HipConv1x1 / HipConvNxN / depthwise / pool, fwd + autograd backward).
Each op runs 8 times on identical inputs; any bitwise difference in
outputs or grads is a nondeterminism (or race) hit.

python benchmarks/conv_det_probe.py
"""

import sys

import torch

sys.path.insert(0, ".")
from adanet_amd.ops.conv import (HipConv1x1, HipConvNxN,  # noqa: E402
                                 HipDepthwiseConv2d, HipPool2d)

dev = "cuda:0"
fails = []


def bitrep(tag, build, reps=8):
    outs = []
    for _ in range(reps):
        torch.manual_seed(5)
        m, x = build()
        y = m(x)
        if isinstance(y, tuple):
            y = y[-1]
        g = torch.ones_like(y)
        y.backward(g)
        snap = [y.detach().clone()]
        snap += [p.grad.detach().clone() for p in m.parameters()
                 if p.grad is not None]
        snap.append(x.grad.detach().clone())
        outs.append(snap)
    ok = all(all(torch.equal(a, b) for a, b in zip(outs[0], o))
             for o in outs[1:])
    print(tag, "OK" if ok else "NONDETERMINISTIC")
    if not ok:
        fails.append(tag)


def mk_c1x1(B, Ci, Co, H):
    def build():
        m = HipConv1x1(Ci, Co, bias=False).cuda().to(torch.bfloat16)
        x = (torch.randn(B, Ci, H, H, device=dev) / 4).to(
            torch.bfloat16).requires_grad_(True)
        return m, x
    return build


def mk_cnxn(B, Ci, Co, H, K, stride):
    def build():
        m = HipConvNxN(Ci, Co, K, stride=stride, padding=K // 2,
                       bias=False).cuda().to(torch.bfloat16)
        x = (torch.randn(B, Ci, H, H, device=dev) / 4).to(
            torch.bfloat16).requires_grad_(True)
        return m, x
    return build


def mk_dw(B, C, H, K, stride):
    def build():
        m = HipDepthwiseConv2d(C, K, stride=stride,
                               padding=K // 2).cuda().to(torch.bfloat16)
        x = (torch.randn(B, C, H, H, device=dev) / 4).to(
            torch.bfloat16).requires_grad_(True)
        return m, x
    return build


def mk_pool(B, C, H, kind, stride):
    def build():
        m = HipPool2d(kind, stride)
        x = (torch.randn(B, C, H, H, device=dev) / 4).to(
            torch.bfloat16).requires_grad_(True)
        return m, x
    return build


def main():
    for (B, Ci, Co, H) in ((64, 32, 32, 32), (64, 96, 32, 32),
                           (64, 160, 64, 16), (256, 320, 64, 8)):
        bitrep("conv1x1 %s" % str((B, Ci, Co, H)), mk_c1x1(B, Ci, Co, H))
    for (B, Ci, Co, H, K, stride) in ((64, 3, 96, 32, 3, 1),
                                      (64, 96, 32, 32, 1, 1),
                                      (64, 160, 32, 16, 1, 2)):
        bitrep("convNxN %s" % str((B, Ci, Co, H, K, stride)),
               mk_cnxn(B, Ci, Co, H, K, stride))
    for (B, C, H, K, stride) in ((64, 32, 32, 5, 1), (64, 64, 16, 3, 1),
                                 (256, 320, 8, 5, 1), (64, 32, 32, 7, 2)):
        bitrep("depthwise %s" % str((B, C, H, K, stride)),
               mk_dw(B, C, H, K, stride))
    for (B, C, H, kind, stride) in ((64, 32, 32, "avg", 1),
                                    (64, 64, 16, "max", 1),
                                    (256, 320, 8, "avg", 2)):
        bitrep("pool %s" % str((B, C, H, kind, stride)),
               mk_pool(B, C, H, kind, stride))
    print("fails:", fails)
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
