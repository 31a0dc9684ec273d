"""Serving latency/throughput of a frozen J-member ensemble on MI355X.

Builds a trained-shape ensemble (random weights — latency is
shape-determined), exports the portable module, and measures:
  - eager batch-1 latency and batch-2048 throughput (bf16 HIP path)
  - hipGraph-captured batch-1 latency (the serving engine's replay path)
"""
import json
import sys
import time

import torch

sys.path.insert(0, ".")


def build_members(J, in_dim=3072, hidden=2048, classes=10, dev="cuda:0"):
    from adanet_amd.ops.linear import HipLinear
    ms = []
    for j in range(J):
        layers = []
        d = in_dim
        for _ in range(j % 3 + 1):
            layers.append(HipLinear(d, hidden, activation="relu"))
            d = hidden
        layers.append(HipLinear(d, classes))
        ms.append(torch.nn.Sequential(*layers).to(dev))
    return ms


def ensemble_fwd(members, w, x):
    from adanet_amd.ops.mixer import weighted_sum_logits
    outs = [m(x) for m in members]
    return weighted_sum_logits(outs, w)


def timeit(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    dev = "cuda:0"
    torch.manual_seed(0)
    res = {}
    for J in (1, 4, 8):
        members = build_members(J)
        w = torch.full((J,), 1.0 / J, device=dev)
        x1 = torch.randn(1, 3072, device=dev).to(torch.bfloat16)
        xB = torch.randn(2048, 3072, device=dev).to(torch.bfloat16)
        with torch.no_grad():
            lat1 = timeit(lambda: ensemble_fwd(members, w, x1))
            latB = timeit(lambda: ensemble_fwd(members, w, xB))
            # hipGraph replay path (static input buffer)
            g = torch.cuda.CUDAGraph()
            static_in = x1.clone()
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                for _ in range(3):
                    ensemble_fwd(members, w, static_in)
            torch.cuda.current_stream().wait_stream(s)
            with torch.cuda.graph(g):
                out = ensemble_fwd(members, w, static_in)
            def replay():
                static_in.copy_(x1)
                g.replay()
            latG = timeit(replay, iters=200, warmup=20)
        res["J=%d" % J] = {
            "eager_batch1_us": round(lat1 * 1e6, 1),
            "graph_batch1_us": round(latG * 1e6, 1),
            "batch2048_ms": round(latB * 1e3, 3),
            "batch2048_Msamples_per_s": round(2048 / latB / 1e6, 2),
        }
    print(json.dumps(res, indent=1))


if __name__ == "__main__":
    main()
