"""Example: customizing AdaNet — your own Builder, Generator, and ensembler.

The runnable analog of the reference's customizing_adanet tutorial
(reference adanet/examples/tutorials/customizing_adanet.ipynb): a custom
search space mixing a LINEAR candidate with DNN candidates, and switching
the ensembler between the complexity-regularized mixture (learned w_j) and
the uniform-average MeanEnsembler.

Usage: python examples/customizing_adanet.py [--ensembler mean|mixture]
"""

import argparse
import sys

import torch
from torch import nn

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import adanet_amd
from adanet_amd import subnetwork as sn
from adanet_amd.ensemble import (ComplexityRegularizedEnsembler,
                                 MeanEnsembler, MixtureWeightType)
from adanet_amd.head import MultiClassHead
from adanet_amd.ops.linear import HipLinear
from adanet_amd.ops.optim import FusedSGD


class _Net(sn.SubnetworkModule):

    def __init__(self, in_dim, hidden_layers, width, classes):
        super().__init__()
        layers = []
        d = in_dim
        for _ in range(hidden_layers):
            layers.append(HipLinear(d, width, activation="relu"))
            d = width
        self.body = nn.Sequential(*layers) if layers else nn.Identity()
        self.head = HipLinear(d, classes)

    def forward(self, x):
        last = self.body(x)
        return last, self.head(last)


class CustomBuilder(sn.Builder):
    """A candidate at a fixed depth (depth 0 == linear model)."""

    def __init__(self, depth, width, classes, lr):
        self._depth = depth
        self._width = width
        self._classes = classes
        self._lr = lr

    @property
    def name(self):
        return "linear" if self._depth == 0 else "dnn_depth_%d" % self._depth

    def build_subnetwork(self, features, logits_dimension, training=True,
                         labels=None, iteration_step=None, summary=None,
                         previous_ensemble=None):
        module = _Net(features.shape[-1], self._depth, self._width,
                      logits_dimension)
        # complexity r(h) = sqrt(depth+1): linear models are "cheapest",
        # so the mixture's L1 penalty prefers them at equal loss.
        return sn.Subnetwork(module=module,
                             complexity=float((self._depth + 1) ** 0.5))

    def build_optimizer(self, parameters, iteration=0):
        return FusedSGD(parameters, lr=self._lr, momentum=0.9)


class CustomGenerator(sn.Generator):
    """Always proposes {linear, depth-1 DNN, depth-2 DNN}."""

    def __init__(self, width=64, classes=10, lr=0.1):
        self._builders = [CustomBuilder(d, width, classes, lr)
                          for d in (0, 1, 2)]

    def generate_candidates(self, previous_ensemble, iteration_number,
                            previous_ensemble_reports, all_reports,
                            config=None):
        return list(self._builders)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--ensembler", choices=["mixture", "mean"],
                   default="mixture")
    p.add_argument("--iterations", type=int, default=3)
    p.add_argument("--model-dir", default="/tmp/adanet_custom")
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

    if args.ensembler == "mixture":
        ensemblers = [ComplexityRegularizedEnsembler(
            mixture_weight_type=MixtureWeightType.SCALAR,
            adanet_lambda=1e-3)]
    else:
        ensemblers = [MeanEnsembler()]

    est = adanet_amd.Estimator(
        head=MultiClassHead(C),
        subnetwork_generator=CustomGenerator(classes=C),
        ensemblers=ensemblers,
        max_iteration_steps=100,
        model_dir=args.model_dir,
        config=adanet_amd.RunConfig(tf_random_seed=1))
    est.train(input_fn, max_steps=100 * args.iterations)
    res = est.evaluate(input_fn, steps=8)
    print("ensembler=%s accuracy=%.3f ensemble=%s" % (
        args.ensembler, res["accuracy"],
        res.get("architecture/adanet/ensembles", "?")))


if __name__ == "__main__":
    main()
