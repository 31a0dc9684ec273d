"""BASELINE config #4: AutoEnsembleEstimator over 8 canned DNN candidates,
one candidate per GPU under round-robin placement (mixture-weight solve
gathers over xGMI at iteration end).

Single-node: python -m torch.distributed.run --nproc-per-node N
             benchmarks/autoensemble_bench.py
Prints one JSON line (same shape as bench.py) from rank 0.
"""
import argparse
import functools
import json
import os
import sys
import tempfile
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=2)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--batch", type=int, default=1024)
    p.add_argument("--train-steps-per-iter", type=int, default=100)
    p.add_argument("--cpu", action="store_true")
    args = p.parse_args()

    import adanet_amd
    from adanet_amd.autoensemble import AutoEnsembleEstimator
    from adanet_amd.distributed import RoundRobinStrategy, comm
    from adanet_amd.head import MultiClassHead
    from adanet_amd.models.canned import DNNEstimator
    from adanet_amd.ops.optim import FusedSGD

    use_gpu = torch.cuda.is_available() and not args.cpu
    comm.maybe_init_process_group()
    world, rank = comm.world_size(), comm.rank()
    device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", rank))) \
        if use_gpu else torch.device("cpu")
    if use_gpu:
        torch.cuda.set_device(device)

    D, C = 784, 10  # MNIST-shape tabular (BASELINE configs #2/#4 family)
    torch.manual_seed(1234)
    teacher = torch.randn(D, C)
    torch.manual_seed(2000 + rank)
    X = torch.randn(args.batch * 8, D)
    Y = (X @ teacher).argmax(dim=1)
    if use_gpu:
        X = X.to(device).to(torch.bfloat16)
        Y = Y.to(device)

    def input_fn():
        def gen():
            i = 0
            while True:
                s = (i * args.batch) % X.shape[0]
                xb = X[s:s + args.batch]
                xb.adanet_cache_key = ("ae", rank, s)
                yield xb, Y[s:s + args.batch]
                i += 1
        return gen()

    head = MultiClassHead(C)
    # 8 canned DNN candidates of varying width/depth — one per GPU at N=8.
    pool = {
        "dnn_%d" % i: DNNEstimator(
            head=head,
            hidden_units=[256 * (1 + i % 4)] * (1 + i // 4),
            optimizer=functools.partial(FusedSGD, lr=0.05, momentum=0.9),
            seed=100 + i)
        for i in range(8)
    }
    est = AutoEnsembleEstimator(
        head=head, candidate_pool=pool,
        max_iteration_steps=args.train_steps_per_iter,
        force_grow=True,
        model_dir=tempfile.mkdtemp(prefix="ae_bench_"),
        config=adanet_amd.RunConfig(tf_random_seed=7, device=str(device),
                                    log_step_count_steps=10 ** 9),
        experimental_placement_strategy=RoundRobinStrategy())

    def one_iteration():
        before = est.iteration_number
        est.train(input_fn, steps=args.train_steps_per_iter)
        assert est.iteration_number == before + 1

    def sync():
        if use_gpu:
            torch.cuda.synchronize(device)
        comm.barrier()

    for _ in range(args.warmup):
        one_iteration()
    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        one_iteration()
    sync()
    elapsed = time.perf_counter() - t0
    if comm.is_initialized():
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_gpu else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())
    res = est.evaluate(input_fn, steps=4)
    if rank == 0:
        print(json.dumps({
            "metric": "adanet_iterations_per_hour",
            "value": args.steps / elapsed * 3600.0,
            "unit": "iterations/hour", "n_gpus": world,
            "steps": args.steps, "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True, "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_gpu else "fp32", "data": "synthetic",
            "config": {"model": "autoensemble-8-canned-dnn (BASELINE #4)",
                       "global_batch": args.batch * world,
                       "candidates_per_iter": 8,
                       "parallelism": "round_robin%d" % world,
                       "final_ensemble_accuracy":
                           float(res.get("accuracy", float("nan")))},
        }))
    if comm.is_initialized():
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
