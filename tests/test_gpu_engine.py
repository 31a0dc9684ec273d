"""End-to-end estimator lifecycle on a single MI355X."""

import json
import os

import pytest
import torch

pytestmark = pytest.mark.gpu


def test_gpu_lifecycle(tmp_path):
    import adanet_amd
    from adanet_amd.head import MultiClassHead
    from adanet_amd.models import simple_dnn

    torch.manual_seed(0)
    N, D, C = 2048, 64, 10
    X = torch.randn(N, D)
    W = torch.randn(D, C)
    Y = (X @ W).argmax(dim=1)

    def input_fn():
        def gen():
            g = torch.Generator().manual_seed(7)
            while True:
                idx = torch.randint(0, N, (256,), generator=g)
                yield X[idx], Y[idx]

        return gen()

    import functools

    from adanet_amd.ops.optim import FusedSGD

    md = str(tmp_path / "model")
    est = adanet_amd.Estimator(
        head=MultiClassHead(C),
        subnetwork_generator=simple_dnn.Generator(
            optimizer_fn=functools.partial(FusedSGD, lr=0.2, momentum=0.9),
            layer_size=256, learn_mixture_weights=True),
        max_iteration_steps=40,
        evaluator=adanet_amd.Evaluator(input_fn=input_fn, steps=4),
        model_dir=md,
        config=adanet_amd.RunConfig(tf_random_seed=42),
    )
    est.train(input_fn, max_steps=120)
    assert est.iteration_number == 3
    res = est.evaluate(input_fn, steps=8)
    # CPU fp32 reference of the identical config reaches ~0.98.
    assert res["accuracy"] > 0.8, res
    arch = json.loads(res["architecture/adanet/ensembles"])
    assert len(arch["subnetworks"]) >= 1
    preds = list(est.predict(lambda: iter([(X[:4], None)])))
    assert len(preds) == 4


def test_gpu_autoensemble(tmp_path):
    import adanet_amd
    from adanet_amd.head import MultiClassHead
    from adanet_amd.models.canned import DNNEstimator, LinearEstimator

    torch.manual_seed(0)
    N, D, C = 1024, 32, 4
    X = torch.randn(N, D)
    Y = (X @ torch.randn(D, C)).argmax(dim=1)

    def input_fn():
        def gen():
            g = torch.Generator().manual_seed(3)
            while True:
                idx = torch.randint(0, N, (128,), generator=g)
                yield X[idx], Y[idx]

        return gen()

    head = MultiClassHead(C)
    est = adanet_amd.AutoEnsembleEstimator(
        head=head,
        candidate_pool={
            "linear": LinearEstimator(head),
            "dnn": DNNEstimator(head, hidden_units=[64, 32]),
        },
        max_iteration_steps=15,
        model_dir=str(tmp_path / "m"),
        config=adanet_amd.RunConfig(tf_random_seed=1),
    )
    est.train(input_fn, max_steps=30)
    assert est.iteration_number == 2
    res = est.evaluate(input_fn, steps=5)
    assert "accuracy" in res


def test_gpu_improve_nas_conv_search(tmp_path):
    """NASNet conv candidates on MI355X: bf16 convs (MIOpen), BN fp32
    stats, our classifier/loss/optimizer kernels, KD hook."""
    import adanet_amd
    from adanet_amd.head import MultiClassHead
    from adanet_amd.models import improve_nas
    from adanet_amd.models.cifar import FakeImageProvider

    hp = improve_nas.Hparams(num_cells=3, num_conv_filters=8,
                             train_steps=6, drop_path_keep=1.0)
    provider = FakeImageProvider(n_examples=128, batch_size=32, seed=3,
                                 augment=False)
    input_fn = provider.get_input_fn()
    est = adanet_amd.Estimator(
        head=MultiClassHead(10, label_smoothing=hp.label_smoothing),
        subnetwork_generator=improve_nas.DynamicGenerator(hp, seed=0),
        max_iteration_steps=6,
        force_grow=True,
        model_dir=str(tmp_path / "m"),
        config=adanet_amd.RunConfig(tf_random_seed=1),
    )
    est.train(input_fn, max_steps=12)
    assert est.iteration_number == 2
    res = est.evaluate(input_fn, steps=2)
    assert res["loss"] == res["loss"]  # finite


def test_gpu_modelflow(tmp_path):
    from adanet_amd.experimental import (AutoEnsemblePhase, GrowStrategy,
                                         InputPhase, MeanEnsemble,
                                         ModelSearch, SequentialController,
                                         TrainerPhase)
    from adanet_amd.ops.linear import HipLinear
    from torch import nn

    torch.manual_seed(0)
    W = torch.randn(16, 3)
    train = []
    g = torch.Generator().manual_seed(0)
    for _ in range(4):
        x = torch.randn(64, 16, generator=g)
        train.append((x, (x @ W).argmax(1)))
    phases = [
        InputPhase(train, train),
        TrainerPhase([
            nn.Sequential(HipLinear(16, 32, activation="relu"),
                          HipLinear(32, 3)) for _ in range(2)
        ], epochs=2),
        AutoEnsemblePhase(ensemblers=[MeanEnsemble],
                          ensemble_strategies=[GrowStrategy()],
                          num_candidates=2),
    ]
    search = ModelSearch(SequentialController(phases))
    search.run()
    best = search.get_best_models(1)[0]
    assert isinstance(best.module, MeanEnsemble)


def test_device_loader_prefetch(tmp_path):
    """DeviceLoader: overlapped pinned H2D feeding a short training run."""
    import adanet_amd
    from adanet_amd.data import DeviceLoader
    from adanet_amd.head import MultiClassHead
    from adanet_amd.models import simple_dnn

    torch.manual_seed(0)
    N, D, C = 1024, 64, 4
    X = torch.randn(N, D)
    Y = (X @ torch.randn(D, C)).argmax(dim=1)

    def cpu_input_fn():
        def gen():
            g = torch.Generator().manual_seed(3)
            while True:
                idx = torch.randint(0, N, (128,), generator=g)
                yield X[idx], Y[idx]

        return gen()

    loader = DeviceLoader(cpu_input_fn, device="cuda:0")
    f, l = next(loader())
    assert f.is_cuda and f.dtype == torch.bfloat16
    assert l.is_cuda

    est = adanet_amd.Estimator(
        head=MultiClassHead(C),
        subnetwork_generator=simple_dnn.Generator(layer_size=32),
        max_iteration_steps=10, model_dir=str(tmp_path / "m"),
        config=adanet_amd.RunConfig(tf_random_seed=1))
    est.train(loader, max_steps=20)
    assert est.iteration_number == 2
