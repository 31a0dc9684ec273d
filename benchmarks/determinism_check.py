"""Refcheck + bit-repeatability + perf for the deterministic two-stage
reductions (relu_bwd_colsum, colsum_bf16 accum, mixer dW).

These three used fp32 atomicAdd with run-dependent ordering: last-ulp
bias/mixer-grad wobble flipped near-tie candidate selections between
otherwise-identical runs (headline swings 9.7k-14.3k iters/hour came
partly from this). Now: per-chunk workspace partials + fixed-order
reducer. This script proves (a) numerics vs torch fp32, (b) bitwise
repeatability across reps, (c) kernel perf did not regress
(relu_bwd_colsum was ~18.4 us @ 2048x2048 with atomics).

python benchmarks/determinism_check.py [--out gpurun_out/det.json]
"""

import argparse
import json
import sys
import time

import torch

sys.path.insert(0, ".")
from adanet_amd.ops import _extension  # noqa: E402

ext = _extension.require()
dev = "cuda:0"


def bench_fn(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6  # us


def check_relu_bwd_colsum(B, C, reps=6):
    torch.manual_seed(1000 + B + C)
    dy = (torch.randn(B, C, device=dev) / 4).to(torch.bfloat16)
    y = (torch.randn(B, C, device=dev)).to(torch.bfloat16)
    db0 = torch.randn(C, device=dev)  # pre-existing arena grad
    outs = []
    for _ in range(reps):
        dz = torch.empty_like(dy)
        db = db0.clone()
        ext.relu_bwd_colsum(dy, y, dz, db, 1.0)
        outs.append((dz.clone(), db.clone()))
    bit_ok = all(torch.equal(outs[0][0], o[0]) and torch.equal(outs[0][1], o[1])
                 for o in outs[1:])
    ref_dz = torch.where(y.float() > 0, dy.float(), torch.zeros(())).to(
        torch.bfloat16)
    ref_db = db0 + ref_dz.float().sum(0)
    num_ok = torch.equal(outs[0][0], ref_dz) and \
        (outs[0][1] - ref_db).abs().max().item() < 0.05 * ref_db.abs().max().item()
    return bit_ok, num_ok


def check_colsum(B, C, accum, reps=6):
    torch.manual_seed(2000 + B + C + accum)
    x = (torch.randn(B, C, device=dev) / 4).to(torch.bfloat16)
    o0 = torch.randn(C, device=dev)
    outs = []
    for _ in range(reps):
        out = o0.clone() if accum else torch.empty(C, device=dev)
        ext.colsum_bf16(x, out, accum)
        outs.append(out.clone())
    bit_ok = all(torch.equal(outs[0], o) for o in outs[1:])
    ref = x.float().sum(0) + (o0 if accum else 0)
    num_ok = (outs[0] - ref).abs().max().item() < 0.05 * (
        ref.abs().max().item() + 1e-6)
    return bit_ok, num_ok


def check_mixer_dw(J, B, C, vector, reps=6):
    torch.manual_seed(3000 + J + B + C)
    members = [(torch.randn(B, C, device=dev) / 4).to(torch.bfloat16)
               for _ in range(J)]
    dY = (torch.randn(B, C, device=dev) / 4).to(torch.bfloat16)
    shape = (J, C) if vector else (J,)
    dw0 = torch.randn(*shape, device=dev).reshape(-1).contiguous()
    outs = []
    for _ in range(reps):
        dw = dw0.clone()
        ext.mixer_bwd_dw_direct(members, dY, dw, int(vector))
        outs.append(dw.clone())
    bit_ok = all(torch.equal(outs[0], o) for o in outs[1:])
    prods = torch.stack([(dY.float() * m.float()) for m in members])
    if vector:
        ref = dw0 + prods.sum(1).reshape(-1)
    else:
        ref = dw0 + prods.sum((1, 2))
    num_ok = (outs[0] - ref).abs().max().item() < 0.05 * (
        ref.abs().max().item() + 1e-6)
    return bit_ok, num_ok


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--out", default=None)
    args = ap.parse_args()
    res = {"checks": [], "perf": {}}
    fails = 0

    for (B, C) in ((2048, 2048), (2048, 3072), (512, 10), (2048, 2049),
                   (64, 2048)):
        bit, num = check_relu_bwd_colsum(B, C)
        res["checks"].append({"op": "relu_bwd_colsum", "B": B, "C": C,
                              "bit": bit, "num": num})
        if not (bit and num):
            fails += 1
            print("FAIL relu_bwd_colsum", B, C, bit, num)
    for accum in (0, 1):
        for (B, C) in ((2048, 2048), (4096, 10), (7, 300)):
            bit, num = check_colsum(B, C, accum)
            res["checks"].append({"op": "colsum", "B": B, "C": C,
                                  "accum": accum, "bit": bit, "num": num})
            if not (bit and num):
                fails += 1
                print("FAIL colsum", B, C, accum, bit, num)
    for vector in (False, True):
        for (J, B, C) in ((3, 2048, 10), (8, 2048, 10), (5, 512, 100)):
            bit, num = check_mixer_dw(J, B, C, vector)
            res["checks"].append({"op": "mixer_dw", "J": J, "B": B, "C": C,
                                  "vector": vector, "bit": bit, "num": num})
            if not (bit and num):
                fails += 1
                print("FAIL mixer_dw", J, B, C, vector, bit, num)
    print("check fails:", fails)

    # perf (prior atomic version: relu_bwd_colsum ~18.4us @ 2048^2)
    dy = torch.randn(2048, 2048, device=dev).to(torch.bfloat16)
    y = torch.randn(2048, 2048, device=dev).to(torch.bfloat16)
    dz = torch.empty_like(dy)
    db = torch.zeros(2048, device=dev)
    res["perf"]["relu_bwd_colsum_2048x2048_us"] = round(bench_fn(
        lambda: ext.relu_bwd_colsum(dy, y, dz, db, 1.0)), 2)
    x = torch.randn(2048, 2048, device=dev).to(torch.bfloat16)
    out = torch.zeros(2048, device=dev)
    res["perf"]["colsum_accum_2048x2048_us"] = round(bench_fn(
        lambda: ext.colsum_bf16(x, out, 1)), 2)
    members = [torch.randn(2048, 10, device=dev).to(torch.bfloat16)
               for _ in range(8)]
    dY = torch.randn(2048, 10, device=dev).to(torch.bfloat16)
    dw = torch.zeros(8, device=dev)
    res["perf"]["mixer_dw_scalar_8x2048x10_us"] = round(bench_fn(
        lambda: ext.mixer_bwd_dw_direct(members, dY, dw, 0)), 2)
    print(json.dumps(res["perf"]))
    if args.out:
        json.dump(res, open(args.out, "w"), indent=1)
    sys.exit(1 if fails else 0)


if __name__ == "__main__":
    main()
