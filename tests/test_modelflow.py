"""ModelFlow e2e on synthetic data (reference test model:
adanet/experimental/keras/model_search_test.py:55-160 — 3 e2e pipelines)."""

import functools

import pytest
import torch
from torch import nn

from adanet_amd.experimental import (AllStrategy, AutoEnsemblePhase,
                                     CompiledModel, GrowStrategy,
                                     InMemoryStorage, InputPhase,
                                     MeanEnsemble, ModelContainer,
                                     ModelSearch, RandomKStrategy,
                                     RepeatPhase, SequentialController,
                                     TrainerPhase, TunerPhase,
                                     WeightedEnsemble)
from adanet_amd.ops.linear import HipLinear


def _dataset(n_batches=4, batch=32, d=8, c=3, seed=0):
    g = torch.Generator().manual_seed(seed)
    W = torch.randn(d, c, generator=g)
    out = []
    for _ in range(n_batches):
        x = torch.randn(batch, d, generator=g)
        out.append((x, (x @ W).argmax(dim=1)))
    return out


def _mlp(d=8, c=3, hidden=16):
    return nn.Sequential(HipLinear(d, hidden, activation="relu",
                                   dtype=torch.float32),
                         HipLinear(hidden, c, dtype=torch.float32))


def test_storage_heap_order():
    s = InMemoryStorage()
    s.save_model(ModelContainer(0.5, "b"))
    s.save_model(ModelContainer(0.1, "a"))
    s.save_model(ModelContainer(0.9, "c"))
    assert s.get_best_models(2) == ["a", "b"]
    assert len(s.get_models()) == 3


def test_trainer_phase_pipeline():
    train, ev = _dataset(seed=0), _dataset(seed=1)
    phases = [
        InputPhase(train, ev),
        TrainerPhase([_mlp(), _mlp()], epochs=2),
    ]
    search = ModelSearch(SequentialController(phases))
    search.run()
    best = search.get_best_models(1)
    assert len(best) == 1
    assert isinstance(best[0], CompiledModel)


def test_autoensemble_phase_pipeline():
    train, ev = _dataset(seed=0), _dataset(seed=1)
    phases = [
        InputPhase(train, ev),
        TrainerPhase([_mlp(), _mlp(), _mlp()], epochs=1),
        AutoEnsemblePhase(
            ensemblers=[MeanEnsemble],
            ensemble_strategies=[GrowStrategy(), AllStrategy(),
                                 RandomKStrategy(2, seed=3)],
            num_candidates=3),
    ]
    search = ModelSearch(SequentialController(phases))
    search.run()
    best = search.get_best_models(1)[0]
    assert isinstance(best.module, MeanEnsemble)


def test_weighted_ensemble_trains():
    train, ev = _dataset(seed=0), _dataset(seed=1)
    subs = [_mlp(), _mlp()]
    ens = WeightedEnsemble(subs, output_units=3)
    model = CompiledModel(ens)
    model.fit(train, epochs=2)
    metrics = model.evaluate(ev)
    assert "loss" in metrics
    # submodels frozen; dense is trainable
    assert all(not p.requires_grad for m in subs for p in m.parameters())
    assert all(p.requires_grad for p in ens.dense.parameters())


def test_tuner_phase_samples_trials():
    train, ev = _dataset(seed=0), _dataset(seed=1)
    phases = [
        InputPhase(train, ev),
        TunerPhase(build_fn=lambda hp: _mlp(hidden=hp["hidden"]),
                   hparam_space={"hidden": [8, 16, 32]}, num_trials=3,
                   epochs=1),
    ]
    search = ModelSearch(SequentialController(phases))
    search.run()
    assert len(search.get_best_models(3)) == 3


def test_repeat_phase():
    train, ev = _dataset(seed=0), _dataset(seed=1)
    storage = InMemoryStorage()
    phases = [
        InputPhase(train, ev),
        RepeatPhase(
            [functools.partial(TrainerPhase, [_mlp()], epochs=1,
                               storage=storage)],
            repetitions=3),
    ]
    search = ModelSearch(SequentialController(phases))
    search.run()
    assert len(storage.get_models()) == 3


def test_threaded_scheduler_phase_barriers():
    """ThreadedScheduler: phase-parallel units with phase barriers
    (AutoEnsemblePhase must see TrainerPhase's completed storage)."""
    from adanet_amd.experimental import ThreadedScheduler
    train, ev = _dataset(seed=0), _dataset(seed=1)
    phases = [
        InputPhase(train, ev),
        TrainerPhase([_mlp(), _mlp(), _mlp()], epochs=1,
                     devices=["cpu", "cpu"]),
        AutoEnsemblePhase(ensemblers=[MeanEnsemble],
                          ensemble_strategies=[AllStrategy()],
                          num_candidates=3),
    ]
    search = ModelSearch(SequentialController(phases),
                         scheduler=ThreadedScheduler(max_workers=3))
    search.run()
    best = search.get_best_models(1)[0]
    assert isinstance(best.module, MeanEnsemble)
    assert len(best.module.submodels) == 3


def test_threaded_scheduler_repeat_phase_serial():
    from adanet_amd.experimental import ThreadedScheduler
    train, ev = _dataset(seed=0), _dataset(seed=1)
    order = []

    class _Tracking(TrainerPhase):

        def work_units(self):
            for wu in super().work_units():
                order.append("unit")
                yield wu

    storage = InMemoryStorage()
    phases = [
        InputPhase(train, ev),
        RepeatPhase([functools.partial(_Tracking, [_mlp()], epochs=1,
                                       storage=storage)], repetitions=3),
    ]
    ModelSearch(SequentialController(phases),
                scheduler=ThreadedScheduler()).run()
    assert len(storage.get_models()) == 3


def test_multigpu_scheduler_cpu_fallback_and_routing():
    from adanet_amd.experimental.schedulers import MultiGpuScheduler

    class _WU:
        def __init__(self, device=None):
            self.device = device
            self.ran = 0

        def execute(self):
            self.ran += 1

    # CPU: n_gpus resolves to 0 -> serial fallback
    sched = MultiGpuScheduler()
    units = [_WU(), _WU("cuda:1"), _WU()]
    sched.schedule(iter(units))
    assert all(u.ran == 1 for u in units)
    # phased with parallel_ok falls back serially too
    units2 = [_WU(), _WU()]
    sched.schedule_phased([(iter(units2), True)])
    assert all(u.ran == 1 for u in units2)
