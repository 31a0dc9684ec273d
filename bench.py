"""bench.py — AdaNet iterations/hour on the CIFAR-10 DNN search space.

The BASELINE.json headline: "AdaNet iterations/hour + final ensemble eval
accuracy, CIFAR-10 DNN search at 1/2/4/8 MI355X". One bench *step* is one
FULL AdaNet boosting iteration (the paper's unit of work): train the
candidate subnetworks simultaneously for --train-steps-per-iter optimizer
steps each, run the Evaluator over the shared eval batches, select the
winner on the complexity-regularized objective, freeze it, checkpoint, and
grow. Data is synthetic CIFAR-10-shaped (3072-dim inputs, 10 classes,
teacher-generated labels), weights random-init, compute dtype bf16.

Multi-GPU (launched by the driver via torch.distributed.run): round-robin
candidate-per-GPU placement (the north star's scaling axis) with the
candidate pool widened by independent restarts so every GPU owns 2
candidates at every N — per-GPU work constant (weak scaling), hipGraphs
stay enabled at all N (no in-step collectives), and the winner broadcasts
over RCCL/xGMI at each iteration end. --placement replication selects
synchronous data-parallel instead (flat per-candidate gradient buckets).
"""

import argparse
import json
import os
import shutil
import sys
import tempfile
import time

import torch


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1,
                   help="world size; when run without a torchrun env and "
                        "N>1, bench.py self-launches under "
                        "torch.distributed.run so a plain `python bench.py "
                        "--gpus 8` measures 8 GPUs")
    p.add_argument("--steps", type=int, default=3,
                   help="AdaNet iterations to time")
    p.add_argument("--warmup", type=int, default=1,
                   help="untimed AdaNet iterations first")
    p.add_argument("--batch", type=int, default=2048, help="per-GPU batch")
    p.add_argument("--hidden", type=int, default=2048)
    p.add_argument("--train-steps-per-iter", type=int, default=150)
    p.add_argument("--eval-batches", type=int, default=8)
    p.add_argument("--restarts", type=int, default=0,
                   help="random restarts per candidate depth (0 = one per "
                        "GPU so round-robin owns 2 candidates per rank)")
    p.add_argument("--placement", choices=["replication", "round_robin"],
                   default="round_robin")
    p.add_argument("--dropout", type=float, default=0.0,
                   help="per-layer dropout (fused into the GEMM relu "
                        "epilogue; the paper space trains with dropout)")
    p.add_argument("--lr", type=float, default=0.05)
    p.add_argument("--model-dir", default=None)
    p.add_argument("--cpu", action="store_true", help="debug on CPU")
    p.add_argument("--per-iter-json", default=None,
                   help="write a per-iteration wall-clock curve (ms vs "
                        "ensemble size + winner arch) to this path, rank 0")
    return p.parse_args()


def _self_launch(args):
    """Re-exec under torchrun: one rank per GPU (driver contract parity)."""
    import socket
    import subprocess
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    cmd = [
        sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
        "--nproc-per-node", str(args.gpus), "--master-addr", "127.0.0.1",
        "--master-port", str(port),
        os.path.abspath(__file__),
    ] + sys.argv[1:]
    sys.exit(subprocess.call(cmd))


def main():
    args = parse_args()
    if args.gpus > 1 and "WORLD_SIZE" not in os.environ:
        _self_launch(args)
    if os.environ.get("ADANET_LOG"):
        import logging
        logging.basicConfig(level=os.environ["ADANET_LOG"])
    sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
    import functools

    import adanet_amd
    from adanet_amd.distributed import (ReplicationStrategy,
                                        RoundRobinStrategy, comm)
    from adanet_amd.head import MultiClassHead
    from adanet_amd.models import simple_dnn
    from adanet_amd.ops.optim import FusedSGD

    use_gpu = torch.cuda.is_available() and not args.cpu
    comm.maybe_init_process_group()
    world = comm.world_size()
    if "WORLD_SIZE" in os.environ and world != args.gpus:
        raise SystemExit(
            "bench.py --gpus {} but torchrun world size is {} — the "
            "reported n_gpus would lie".format(args.gpus, world))
    rank = comm.rank()
    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    device = torch.device("cuda", local_rank) if use_gpu else torch.device(
        "cpu")
    if use_gpu:
        torch.cuda.set_device(device)

    D, C = 3072, 10
    torch.manual_seed(1234)  # same teacher everywhere
    teacher = torch.randn(D, C)
    torch.manual_seed(1000 + rank)  # different data shards per rank

    # HBM-resident synthetic dataset: a pool of batches cycled forever.
    n_pool = 32
    pool = []
    for i in range(n_pool):
        x = torch.randn(args.batch, D)
        y = (x @ teacher).argmax(dim=1)
        if use_gpu:
            x = x.to(device).to(torch.bfloat16)
            y = y.to(device)
        # Resident-dataset batch: lets the engine replay frozen-member
        # logits from the HBM cache instead of recomputing per step.
        x.adanet_cache_key = ("train", rank, i)
        pool.append((x, y))
    eval_pool = []
    torch.manual_seed(5000 + rank)
    for i in range(args.eval_batches):
        x = torch.randn(args.batch, D)
        y = (x @ teacher).argmax(dim=1)
        if use_gpu:
            x = x.to(device).to(torch.bfloat16)
            y = y.to(device)
        x.adanet_cache_key = ("eval", rank, i)
        eval_pool.append((x, y))

    def input_fn():
        def gen():
            i = 0
            while True:
                yield pool[i % n_pool]
                i += 1

        return gen()

    def eval_input_fn():
        return iter(list(eval_pool))

    model_dir = args.model_dir or tempfile.mkdtemp(prefix="adanet_bench_")
    placement = (ReplicationStrategy() if args.placement == "replication"
                 else RoundRobinStrategy())

    # Paper-style CIFAR DNN search space: candidates at the current depth
    # and one deeper, width --hidden, fused momentum-SGD. The pool widens
    # with the GPU count (independent restarts) so round-robin placement
    # keeps 2 candidates per GPU at every N — the north star's
    # candidate-per-GPU scaling axis with constant per-GPU work.
    restarts = args.restarts if args.restarts > 0 else world
    generator = simple_dnn.Generator(
        optimizer_fn=functools.partial(FusedSGD, lr=args.lr, momentum=0.9),
        mixture_optimizer_fn=functools.partial(FusedSGD, lr=args.lr / 10),
        layer_size=args.hidden,
        initial_num_layers=1,
        learn_mixture_weights=True,
        dropout=args.dropout,
        seed=77,
        num_restarts=restarts)

    est = adanet_amd.Estimator(
        head=MultiClassHead(C),
        subnetwork_generator=generator,
        max_iteration_steps=args.train_steps_per_iter,
        evaluator=adanet_amd.Evaluator(input_fn=eval_input_fn,
                                       steps=args.eval_batches),
        force_grow=True,
        adanet_lambda=1e-4,
        model_dir=model_dir,
        config=adanet_amd.RunConfig(tf_random_seed=42,
                                    device=str(device),
                                    log_step_count_steps=10 ** 9),
        experimental_placement_strategy=placement,
    )

    def one_iteration():
        before = est.iteration_number
        est.train(input_fn, steps=args.train_steps_per_iter)
        assert est.iteration_number == before + 1, (
            "bench step must complete one full AdaNet iteration")

    def sync():
        if use_gpu:
            torch.cuda.synchronize(device)
        comm.barrier()

    curve = []

    def record_iter(dt_ms, timed):
        t = est.iteration_number - 1
        entry = {"iteration": t, "ms": round(dt_ms, 2), "timed": timed}
        try:
            arch = json.load(
                open(os.path.join(model_dir,
                                  "architecture-%d.json" % t)))
            entry["winner"] = arch["subnetworks"][-1]["builder_name"]
            entry["ensemble_size"] = len(arch["subnetworks"])
        except Exception:
            pass
        entry["phase_secs"] = {k: round(v, 3)
                               for k, v in est._phase_secs.items()}
        curve.append(entry)

    for _ in range(args.warmup):
        ti = time.perf_counter()
        one_iteration()
        if args.per_iter_json:
            sync()
            record_iter((time.perf_counter() - ti) * 1000.0, False)

    sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        ti = time.perf_counter()
        one_iteration()
        if args.per_iter_json:
            sync()
            record_iter((time.perf_counter() - ti) * 1000.0, True)
    sync()
    elapsed = time.perf_counter() - t0

    # MAX over ranks.
    if comm.is_initialized():
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device=device if use_gpu else "cpu")
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = float(t.item())

    res = est.evaluate(eval_input_fn, steps=args.eval_batches)
    final_acc = float(res.get("accuracy", float("nan")))

    iters_per_hour = args.steps / elapsed * 3600.0
    if rank == 0:
        out = {
            "metric": "adanet_iterations_per_hour",
            "value": iters_per_hour,
            "unit": "iterations/hour",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if use_gpu else "fp32",
            "data": "synthetic",
            "config": {
                "model": "cifar10-dnn-search (adanet paper space)",
                "global_batch": args.batch * world,
                "input_dim": D,
                "n_classes": C,
                "hidden": args.hidden,
                "candidates_per_iter": 2 * restarts,
                "train_steps_per_iter": args.train_steps_per_iter,
                "eval_batches": args.eval_batches,
                "dropout": args.dropout,
                "parallelism": ("dp%d" % world
                                if args.placement == "replication" else
                                "round_robin%d" % world),
                "final_ensemble_accuracy": final_acc,
                "final_ensemble_size": est.iteration_number,
            },
        }
        print(json.dumps(out))
        if args.per_iter_json:
            with open(args.per_iter_json, "w") as f:
                json.dump({"curve": curve}, f, indent=1)
    if args.model_dir is None:
        shutil.rmtree(model_dir, ignore_errors=True)
    if comm.is_initialized():
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
