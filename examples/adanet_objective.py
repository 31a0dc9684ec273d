"""Example: the AdaNet objective in action — lambda controls ensemble growth.

Runnable analog of the reference's adanet_objective tutorial
(adanet/examples/tutorials/adanet_objective.ipynb): sweeping the
complexity-penalty strength shows the trade-off the mixture weights make
between fitting the data and adding (deeper = more complex) subnetworks.

Usage: python examples/adanet_objective.py
"""

import functools
import sys

import torch

sys.path.insert(0, __import__("os").path.dirname(
    __import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import adanet_amd
from adanet_amd.head import MultiClassHead
from adanet_amd.models import simple_dnn
from adanet_amd.ops.optim import FusedSGD


def run(lam, input_fn, C):
    est = adanet_amd.Estimator(
        head=MultiClassHead(C),
        subnetwork_generator=simple_dnn.Generator(
            optimizer_fn=functools.partial(FusedSGD, lr=0.1, momentum=0.9),
            layer_size=32, learn_mixture_weights=True, seed=3),
        max_iteration_steps=80,
        adanet_lambda=lam,
        model_dir="/tmp/adanet_objective_%g" % lam,
        config=adanet_amd.RunConfig(tf_random_seed=11))
    est.train(input_fn, max_steps=240)  # 3 boosting iterations
    res = est.evaluate(input_fn, steps=8)
    # materialize the frozen best ensemble: which architectures won?
    X0, _ = next(iter(input_fn()))
    ens, _ = est._rebuild_previous_ensemble(est.iteration_number, X0)
    members = [ws.subnetwork.name.split("_", 1)[1]
               for ws in ens.weighted_subnetworks]
    total_complexity = sum(float(ws.subnetwork.complexity)
                           for ws in ens.weighted_subnetworks)
    return res["accuracy"], members, total_complexity


def main():
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

    print("%-10s %-10s %-12s %s" % ("lambda", "accuracy",
                                      "sum r(h_j)", "selected members"))
    for lam in (0.0, 0.1, 1.0):
        acc, members, cx = run(lam, input_fn, C)
        print("%-10g %-10.3f %-12.2f %s" % (lam, acc, cx, members))
    print("\nLarger lambda steers selection toward LOWER-complexity "
          "subnetworks (r(h) = sqrt(depth)): the complexity-regularized "
          "objective F(w) = L(w) + sum_j (lambda*r_j + beta)|w_j| trades "
          "fit against architecture cost (docs/ALGORITHM.md).")


if __name__ == "__main__":
    main()
