"""Example: NASNet CIFAR conv search (improve_nas analog) on fake data.

The trainer CLI analog of research/improve_nas/trainer/trainer.py.
Usage: python examples/train_improve_nas.py [--boosting-iterations 2]
"""

import argparse
import sys

sys.path.insert(0, ".")
import adanet_amd
from adanet_amd.head import MultiClassHead
from adanet_amd.models import improve_nas
from adanet_amd.models.cifar import FakeImageProvider


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--boosting-iterations", type=int, default=2)
    p.add_argument("--train-steps", type=int, default=50)
    p.add_argument("--num-cells", type=int, default=3)
    p.add_argument("--num-conv-filters", type=int, default=10)
    p.add_argument("--knowledge-distillation", default="adaptive",
                   choices=["none", "adaptive", "born_again"])
    p.add_argument("--batch-size", type=int, default=64)
    p.add_argument("--model-dir", default="/tmp/adanet_improve_nas")
    args = p.parse_args()

    hp = improve_nas.Hparams(
        num_cells=args.num_cells,
        num_conv_filters=args.num_conv_filters,
        knowledge_distillation=args.knowledge_distillation,
        boosting_iterations=args.boosting_iterations,
        train_steps=args.train_steps)
    provider = FakeImageProvider(n_classes=10, n_examples=512,
                                 batch_size=args.batch_size, seed=1)
    input_fn = provider.get_input_fn()

    estimator = adanet_amd.Estimator(
        head=MultiClassHead(10, label_smoothing=hp.label_smoothing),
        subnetwork_generator=improve_nas.DynamicGenerator(hp, seed=0),
        max_iteration_steps=args.train_steps,
        force_grow=hp.force_grow,
        max_iterations=args.boosting_iterations,
        model_dir=args.model_dir,
        config=adanet_amd.RunConfig(tf_random_seed=1),
    )
    estimator.train(
        input_fn, max_steps=args.boosting_iterations * args.train_steps)
    results = estimator.evaluate(input_fn, steps=8)
    print("accuracy: %.4f  loss: %.4f" % (results["accuracy"], results["loss"]))


if __name__ == "__main__":
    main()
