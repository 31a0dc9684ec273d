"""Example: AdaNet DNN search on synthetic data (1 GPU or CPU).

Usage: python examples/train_simple_dnn.py [--iterations 3]
"""

import argparse
import functools
import sys

import torch

sys.path.insert(0, ".")
import adanet_amd
from adanet_amd.head import MultiClassHead
from adanet_amd.models import simple_dnn
from adanet_amd.ops.optim import FusedSGD


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--iterations", type=int, default=3)
    p.add_argument("--steps-per-iteration", type=int, default=100)
    p.add_argument("--model-dir", default="/tmp/adanet_simple_dnn")
    args = p.parse_args()

    torch.manual_seed(0)
    N, D, C = 4096, 64, 10
    X = torch.randn(N, D)
    Y = (X @ torch.randn(D, C)).argmax(dim=1)

    def input_fn():
        def gen():
            g = torch.Generator().manual_seed(7)
            while True:
                idx = torch.randint(0, N, (256,), generator=g)
                yield X[idx], Y[idx]
        return gen()

    estimator = adanet_amd.Estimator(
        head=MultiClassHead(C),
        subnetwork_generator=simple_dnn.Generator(
            optimizer_fn=functools.partial(FusedSGD, lr=0.1, momentum=0.9),
            layer_size=256, learn_mixture_weights=True),
        max_iteration_steps=args.steps_per_iteration,
        evaluator=adanet_amd.Evaluator(input_fn=input_fn, steps=8),
        force_grow=True,
        max_iterations=args.iterations,
        model_dir=args.model_dir,
        config=adanet_amd.RunConfig(tf_random_seed=42),
    )
    estimator.train(input_fn, max_steps=args.iterations * args.steps_per_iteration)
    results = estimator.evaluate(input_fn, steps=16)
    print("accuracy: %.4f  loss: %.4f" % (results["accuracy"], results["loss"]))
    print("architecture:", results["architecture/adanet/ensembles"])


if __name__ == "__main__":
    main()
