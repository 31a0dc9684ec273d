"""Reproduce the streams x NASNet nondeterminism outside the estimator:
two NasNetCIFAR candidates trained in the engine's exact stream pattern
(per-candidate HIP stream, default-stream event at step start, device
join at step end), repeated with identical seeds; parameter bytes
compared across reps.

python benchmarks/streams_det_probe.py [--steps 8] [--reps 4]
"""

import argparse
import hashlib
import sys

import torch

sys.path.insert(0, ".")
from adanet_amd.models.nasnet import NasNetCIFAR  # noqa: E402


def run_once(steps, use_streams, keep, engine_opt=False, engine_full=False):
    torch.manual_seed(7)
    models = [NasNetCIFAR(num_cells=3, num_conv_filters=32,
                          drop_path_keep=keep).cuda(),
              NasNetCIFAR(num_cells=4, num_conv_filters=32,
                          drop_path_keep=keep).cuda()]
    from adanet_amd.ops.linear import restore_fp32_params
    for m in models:
        m.to(torch.bfloat16)
        restore_fp32_params(m)
        m.train()
    if engine_opt:
        from adanet_amd.ops.optim import CosineLR, FusedSGD
        opts = [FusedSGD(list(m.parameters()), lr=0.025, momentum=0.9,
                         weight_decay=5e-4) for m in models]
        scheds = [CosineLR(o, steps) for o in opts]
    else:
        opts = [torch.optim.SGD(m.parameters(), lr=0.025, momentum=0.9)
                for m in models]
        scheds = []
    torch.manual_seed(13)
    x = (torch.randn(64, 3, 32, 32, device="cuda") / 4).to(torch.bfloat16)
    y = torch.randint(0, 10, (64,), device="cuda")
    streams = [torch.cuda.Stream(), torch.cuda.Stream()]
    cur = torch.cuda.current_stream()
    head = mixers = mix_opts = frozen = loss_buf = loss_ctr = None
    if engine_full:
        # engine extras: fused-xent head, per-candidate mixer over
        # [frozen static, own logits], device loss ring.
        from adanet_amd.head import MultiClassHead
        from adanet_amd.ops.mixer import weighted_sum_logits
        from adanet_amd.ops.optim import FusedSGD
        head = MultiClassHead(10)
        frozen = (torch.randn(64, 10, device="cuda") / 4).to(torch.bfloat16)
        mixers = [torch.nn.ParameterDict({
            "w": torch.nn.Parameter(torch.full((2,), 0.5, device="cuda")),
            "b": torch.nn.Parameter(torch.zeros(10, device="cuda"))})
            for _ in models]
        mix_opts = [FusedSGD([m["w"], m["b"]], lr=0.01) for m in mixers]
        loss_buf = torch.full((64, 4), float("nan"), device="cuda")
        loss_ctr = torch.zeros((), device="cuda", dtype=torch.long)
        globals()["_wsl"] = weighted_sum_logits
    for _ in range(steps):
        ev = torch.cuda.Event()
        ev.record()
        for m, o, s in zip(models, opts, streams):
            if use_streams:
                ev.wait(s)
                ctx = torch.cuda.stream(s)
            else:
                import contextlib
                ctx = contextlib.nullcontext()
            with ctx:
                o.zero_grad(set_to_none=True)
                _, logits = m(x)
                loss = torch.nn.functional.cross_entropy(logits.float(), y)
                if engine_opt:
                    from adanet_amd.ops.linear import direct_grad_writes
                    with direct_grad_writes():
                        loss.backward()
                else:
                    loss.backward()
                o.step()
                if engine_full:
                    mi = models.index(m)
                    md, mo = mixers[mi], mix_opts[mi]
                    from adanet_amd.ops.linear import direct_grad_writes
                    wsl = globals()["_wsl"]
                    mixed = wsl([frozen, logits.detach()], md["w"], md["b"])
                    mloss = head.loss(mixed, y) + 1e-3 * md["w"].abs().sum()
                    mo.zero_grad(set_to_none=True)
                    with direct_grad_writes():
                        mloss.backward()
                    mo.step()
        for sc in scheds:
            sc.step()
        if use_streams:
            for s in streams:
                cur.wait_stream(s)
        if engine_full:
            row = torch.stack([m["w"].detach().sum() for m in mixers] +
                              [m["b"].detach().sum() for m in mixers])
            idx = torch.remainder(loss_ctr, 64)
            loss_buf.index_copy_(0, idx.reshape(1), row.reshape(1, 4))
            loss_ctr.add_(1)
    torch.cuda.synchronize()
    h = hashlib.sha256()
    for m in models:
        for name, p in sorted(m.named_parameters()):
            h.update(p.detach().float().cpu().numpy().tobytes())
    if engine_full:
        for md in mixers:
            h.update(md["w"].detach().float().cpu().numpy().tobytes())
            h.update(md["b"].detach().float().cpu().numpy().tobytes())
        h.update(loss_buf.cpu().numpy().tobytes())
    return h.hexdigest()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--reps", type=int, default=4)
    ap.add_argument("--no-streams", action="store_true")
    ap.add_argument("--no-droppath", action="store_true")
    ap.add_argument("--engine-opt", action="store_true",
                    help="FusedSGD + CosineLR + direct-to-arena grads")
    ap.add_argument("--engine-full", action="store_true",
                    help="+ fused-xent head, mixer ensemble loss, loss ring")
    args = ap.parse_args()
    keep = 1.0 if args.no_droppath else 0.9
    ref = run_once(args.steps, not args.no_streams, keep, args.engine_opt,
                   args.engine_full)
    mism = 0
    for _ in range(args.reps):
        if run_once(args.steps, not args.no_streams, keep,
                    args.engine_opt, args.engine_full) != ref:
            mism += 1
    print("mismatches: %d/%d (streams=%s droppath=%s engine_opt=%s full=%s)" %
          (mism, args.reps, not args.no_streams, keep < 1.0,
           args.engine_opt, args.engine_full))
    sys.exit(1 if mism else 0)


if __name__ == "__main__":
    main()
