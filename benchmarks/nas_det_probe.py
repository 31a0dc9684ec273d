"""Bisect NAS-space GPU nondeterminism: train a bare NasNetCIFAR for N
steps twice with identical seeds and compare parameter bytes exactly.
Isolates the model/kernel stack from the estimator machinery (the CPU
run of the full bench is bit-deterministic; the GPU run is not).

python benchmarks/nas_det_probe.py [--steps 10] [--no-droppath]
"""

import argparse
import hashlib
import sys

import torch
import torch.nn.functional as F

sys.path.insert(0, ".")
from adanet_amd.models.nasnet import NasNetCIFAR  # noqa: E402
from adanet_amd.ops import conv as aconv  # noqa: E402


def disable_native(classes):
    """Monkeypatch chosen Hip conv-family modules to their torch fallback
    so the racy class can be bisected inside the full model."""
    if "dw" in classes:
        aconv.HipDepthwiseConv2d.forward = lambda self, x: F.conv2d(
            x, self.weight.to(x.dtype), None, self.stride, self.padding,
            groups=self.channels)
    if "c1" in classes:
        aconv.HipConv1x1.forward = lambda self, x: F.conv2d(
            x, self.weight.to(x.dtype),
            self.bias.to(x.dtype) if self.bias is not None else None)
    if "cn" in classes:
        aconv.HipConvNxN.forward = lambda self, x: F.conv2d(
            x, self.weight.to(x.dtype),
            self.bias.to(x.dtype) if self.bias is not None else None,
            self.stride, self.padding)
    if "bn" in classes:
        from adanet_amd.ops import batchnorm as abn
        def _bn_fwd(self, x):
            return F.batch_norm(
                x.float(), self.running_mean, self.running_var,
                self.weight.float() if self.weight is not None else None,
                self.bias.float() if self.bias is not None else None,
                self.training, self.momentum, self.eps).to(x.dtype)
        abn.HipBatchNorm2d.forward = _bn_fwd
    if "fc" in classes:
        from adanet_amd.ops import linear as alin
        def _fc_fwd(self, x):
            b = self.bias
            return F.linear(x, self.weight.to(x.dtype),
                            b.to(x.dtype) if b is not None else None)
        alin.HipLinear.forward = _fc_fwd
    if "pool" in classes:
        aconv.HipPool2d.forward = lambda self, x: (
            F.max_pool2d(x, 3, self.stride, 1) if self.kind == "max"
            else F.avg_pool2d(x, 3, self.stride, 1,
                              count_include_pad=False))


def run_once(steps, drop_path_keep, dtype=torch.bfloat16):
    torch.manual_seed(7)
    model = NasNetCIFAR(num_cells=3, num_conv_filters=32,
                        drop_path_keep=drop_path_keep).cuda()
    from adanet_amd.ops.linear import restore_fp32_params
    model = model.to(torch.bfloat16)
    restore_fp32_params(model)
    model.train()
    opt = torch.optim.SGD([p for p in model.parameters()], lr=0.025,
                          momentum=0.9)
    torch.manual_seed(13)
    x = (torch.randn(64, 3, 32, 32, device="cuda") / 4).to(dtype)
    y = torch.randint(0, 10, (64,), device="cuda")
    losses = []
    for _ in range(steps):
        opt.zero_grad(set_to_none=True)
        _, logits = model(x)
        loss = torch.nn.functional.cross_entropy(logits.float(), y)
        loss.backward()
        opt.step()
        losses.append(loss.item())
    h = hashlib.sha256()
    for name, p in sorted(model.named_parameters()):
        h.update(name.encode())
        h.update(p.detach().float().cpu().numpy().tobytes())
    return h.hexdigest(), losses


def gradcmp(steps, keep, reps):
    """Run fwd+bwd repeatedly on FIXED weights; report every parameter
    whose gradient bytes differ across reps (pinpoints the racy op)."""
    torch.manual_seed(7)
    model = NasNetCIFAR(num_cells=3, num_conv_filters=32,
                        drop_path_keep=keep).cuda()
    from adanet_amd.ops.linear import restore_fp32_params
    model = model.to(torch.bfloat16)
    restore_fp32_params(model)
    model.train()
    torch.manual_seed(13)
    x0 = (torch.randn(64, 3, 32, 32, device="cuda") / 4).to(torch.bfloat16)
    y = torch.randint(0, 10, (64,), device="cuda")
    ref = None
    bad = {}
    for rep in range(reps):
        for p in model.parameters():
            p.grad = None
        torch.manual_seed(99)  # identical drop_path draws every rep
        x = x0.clone().requires_grad_(True)
        _, logits = model(x)
        loss = torch.nn.functional.cross_entropy(logits.float(), y)
        loss.backward()
        snap = {n: p.grad.detach().clone()
                for n, p in model.named_parameters() if p.grad is not None}
        snap["__x__"] = x.grad.detach().clone()
        snap["__loss__"] = loss.detach().clone()
        if ref is None:
            ref = snap
        else:
            for n, g in snap.items():
                if not torch.equal(ref[n], g):
                    bad.setdefault(n, 0)
                    bad[n] += 1
    print("diverging tensors (%d):" % len(bad))
    for n, c in sorted(bad.items()):
        print("  %s  (%d/%d reps)" % (n, c, reps - 1))
    return len(bad)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--no-droppath", action="store_true")
    ap.add_argument("--disable", default="",
                    help="comma list of dw,c1,cn,pool to force torch fallback")
    ap.add_argument("--reps", type=int, default=1,
                    help="repeat the A/B comparison; races are flaky")
    ap.add_argument("--gradcmp", action="store_true")
    args = ap.parse_args()
    if args.disable:
        disable_native(args.disable.split(","))
    keep = 1.0 if args.no_droppath else 0.9
    if args.gradcmp:
        sys.exit(1 if gradcmp(args.steps, keep, args.reps) else 0)
    mism = 0
    ref_h, ref_l = run_once(args.steps, keep)
    for _ in range(args.reps):
        h2, l2 = run_once(args.steps, keep)
        if h2 != ref_h:
            mism += 1
    print("mismatches: %d/%d" % (mism, args.reps))
    if mism:
        for i, (a, b) in enumerate(zip(ref_l, l2)):
            flag = "  <-- diverges" if a != b else ""
            print("step %d loss: %.9f vs %.9f%s" % (i, a, b, flag))
    sys.exit(1 if mism else 0)


if __name__ == "__main__":
    main()
