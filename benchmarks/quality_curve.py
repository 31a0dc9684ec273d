"""Quality curve: per-iteration best-ensemble eval accuracy (the second half
of the BASELINE metric — "final ensemble eval accuracy").

python benchmarks/quality_curve.py [--iterations 10]
"""
import argparse
import functools
import json
import sys
import tempfile

import torch

sys.path.insert(0, ".")
import adanet_amd
from adanet_amd.head import MultiClassHead
from adanet_amd.models import simple_dnn
from adanet_amd.ops.optim import FusedSGD


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iterations", type=int, default=10)
    p.add_argument("--steps-per-iter", type=int, default=150)
    p.add_argument("--batch", type=int, default=2048)
    p.add_argument("--hidden", type=int, default=2048)
    args = p.parse_args()
    use_gpu = torch.cuda.is_available()
    dev = torch.device("cuda:0") if use_gpu else torch.device("cpu")

    D, C = 3072, 10
    torch.manual_seed(1234)
    teacher = torch.randn(D, C)
    torch.manual_seed(99)
    pool, eval_pool = [], []
    for i in range(24):  # CIFAR-10-sized resident set (24 x 2048 = 49k)
        x = torch.randn(args.batch, D)
        y = (x @ teacher).argmax(1)
        if use_gpu:
            x, y = x.to(dev).to(torch.bfloat16), y.to(dev)
        x.adanet_cache_key = ("train", i)
        pool.append((x, y))
    for i in range(8):
        x = torch.randn(args.batch, D)
        y = (x @ teacher).argmax(1)
        if use_gpu:
            x, y = x.to(dev).to(torch.bfloat16), y.to(dev)
        x.adanet_cache_key = ("eval", i)
        eval_pool.append((x, y))

    def input_fn():
        def gen():
            i = 0
            while True:
                yield pool[i % len(pool)]
                i += 1
        return gen()

    def eval_input_fn():
        return iter(list(eval_pool))

    est = adanet_amd.Estimator(
        head=MultiClassHead(C),
        subnetwork_generator=simple_dnn.Generator(
            optimizer_fn=functools.partial(FusedSGD, lr=0.02, momentum=0.9),
            mixture_optimizer_fn=functools.partial(FusedSGD, lr=0.005),
            layer_size=args.hidden, initial_num_layers=1,
            learn_mixture_weights=True, seed=7),
        max_iteration_steps=args.steps_per_iter,
        evaluator=adanet_amd.Evaluator(input_fn=eval_input_fn, steps=8),
        force_grow=True,
        adanet_lambda=1e-4,
        model_dir=tempfile.mkdtemp(),
        config=adanet_amd.RunConfig(tf_random_seed=42,
                                    log_step_count_steps=10**9),
    )
    curve = []
    for t in range(args.iterations):
        est.train(input_fn, steps=args.steps_per_iter)
        res = est.evaluate(eval_input_fn, steps=8)
        arch = json.loads(res["architecture/adanet/ensembles"])
        curve.append({"iteration": t, "accuracy": round(res["accuracy"], 4),
                      "loss": round(res["loss"], 4),
                      "members": len(arch["subnetworks"])})
        print(json.dumps(curve[-1]))
    print(json.dumps({"curve": curve}))


if __name__ == "__main__":
    main()
