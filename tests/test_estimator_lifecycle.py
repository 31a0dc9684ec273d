"""Estimator integration tests — behavioral invariants on tiny data.

Reference test model: adanet/core/estimator_test.py (lifecycle :1022,
checkpoints :1659, force_grow :3002, replay :3235, NaN :3081) ported as
behavioral invariants (iteration counts, architecture JSON, checkpoint
layout) rather than TF golden losses (SURVEY.md section 7 "hard parts" (e)).
"""

import glob
import json
import math
import os

import pytest
import torch
from torch import nn

import adanet_amd
from adanet_amd import replay
from adanet_amd.head import MultiClassHead
from adanet_amd.models import simple_dnn
from adanet_amd.subnetwork import Builder, SimpleGenerator, Subnetwork


def _make_estimator(model_dir, input_fn, **kwargs):
    defaults = dict(
        head=MultiClassHead(4),
        subnetwork_generator=simple_dnn.Generator(layer_size=8),
        max_iteration_steps=10,
        model_dir=model_dir,
        config=adanet_amd.RunConfig(tf_random_seed=42),
    )
    defaults.update(kwargs)
    return adanet_amd.Estimator(**defaults)


def test_constructor_validation(model_dir):
    with pytest.raises(ValueError):
        adanet_amd.Estimator(head=MultiClassHead(4),
                             subnetwork_generator=None,
                             max_iteration_steps=10)
    with pytest.raises(ValueError):
        _make_estimator(model_dir, None, max_iteration_steps=0)
    with pytest.raises(ValueError):
        _make_estimator(model_dir, None, max_iterations=0)


def test_lifecycle_iterations_and_layout(model_dir, synthetic_classification):
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(model_dir, input_fn)
    est.train(input_fn, max_steps=30)
    # 3 iterations x 10 steps.
    assert est.iteration_number == 3
    assert est.global_step == 30
    for t in range(3):
        path = os.path.join(model_dir, "architecture-{}.json".format(t))
        assert os.path.exists(path), path
        arch = json.loads(open(path).read())
        assert arch["iteration_number"] == t
        assert len(arch["replay_indices"]) == t + 1
    assert os.path.exists(os.path.join(model_dir, "checkpoint"))
    assert glob.glob(os.path.join(model_dir, "increment.ckpt-*.pt"))
    assert os.path.isdir(os.path.join(model_dir, "train_manager", "t0"))


def test_train_steps_vs_max_steps(model_dir, synthetic_classification):
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(model_dir, input_fn)
    est.train(input_fn, steps=10)
    assert est.global_step == 10
    est.train(input_fn, steps=10)
    assert est.global_step == 20
    # max_steps is absolute
    est.train(input_fn, max_steps=25)
    assert est.global_step == 25
    assert est.iteration_number == 2  # 25 steps -> iteration 2 in progress


def test_max_iterations_stops_early(model_dir, synthetic_classification):
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(model_dir, input_fn, max_iterations=2)
    est.train(input_fn, max_steps=100)
    assert est.iteration_number == 2
    assert est.global_step == 20


def test_checkpoint_resume_mid_iteration(model_dir, synthetic_classification):
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(model_dir, input_fn)
    est.train(input_fn, steps=15)  # mid-iteration 1
    assert est.iteration_number == 1
    est2 = _make_estimator(model_dir, input_fn)
    assert est2.iteration_number == 1
    assert est2.global_step == 15
    est2.train(input_fn, steps=5)
    assert est2.iteration_number == 2
    assert est2.global_step == 20


def test_grown_ensemble_members_accumulate(model_dir,
                                           synthetic_classification):
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(model_dir, input_fn, force_grow=True)
    est.train(input_fn, max_steps=30)
    arch = json.loads(
        open(os.path.join(model_dir, "architecture-2.json")).read())
    assert len(arch["subnetworks"]) == 3  # force_grow: one member per round


def test_evaluate_predict_export(model_dir, synthetic_classification):
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(model_dir, input_fn)
    est.train(input_fn, max_steps=20)
    res = est.evaluate(input_fn, steps=5)
    assert "loss" in res and "accuracy" in res
    assert res["accuracy"] >= 0.0
    preds = list(est.predict(lambda: iter([(X[:6], None)])))
    assert len(preds) == 6
    assert "probabilities" in preds[0]
    export_dir = est.export_saved_model(os.path.join(model_dir, "export"))
    assert os.path.exists(os.path.join(export_dir, "saved_model.pt"))
    assert os.path.exists(os.path.join(export_dir, "architecture.json"))


def test_live_prev_matches_checkpoint_rebuild(model_dir,
                                              synthetic_classification):
    """The in-process live-module fast path and a fresh restore from the
    checkpoint must describe the same frozen best ensemble."""
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(model_dir, input_fn)
    est.train(input_fn, max_steps=20)
    live_res = est.evaluate(input_fn, steps=4)
    est2 = _make_estimator(model_dir, input_fn)  # rebuilds from checkpoint
    rebuilt_res = est2.evaluate(input_fn, steps=4)
    assert live_res["loss"] == pytest.approx(rebuilt_res["loss"], abs=1e-5)
    assert (live_res["architecture/adanet/ensembles"] ==
            rebuilt_res["architecture/adanet/ensembles"])


def test_evaluate_before_training_raises(model_dir, synthetic_classification):
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(model_dir, input_fn)
    with pytest.raises(ValueError):
        est.evaluate(input_fn, steps=1)


def test_replay_config_overrides_selection(model_dir,
                                           synthetic_classification):
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(
        model_dir, input_fn,
        replay_config=replay.Config(best_ensemble_indices=[1, 0]))
    est.train(input_fn, max_steps=20)
    arch0 = json.loads(
        open(os.path.join(model_dir, "architecture-0.json")).read())
    # candidate index 1 at iteration 0 is the deeper DNN candidate.
    assert arch0["replay_indices"][0] == 1
    assert "1_layer_dnn" in [
        s["builder_name"] for s in arch0["subnetworks"]
    ]


class _NanBuilder(Builder):
    """Reference _NanLossBuilder (estimator_test.py:226)."""

    @property
    def name(self):
        return "nan"

    def build_subnetwork(self, features, logits_dimension, training,
                         previous_ensemble=None):

        class _M(nn.Module):

            def __init__(self):
                super().__init__()
                self.w = nn.Parameter(torch.ones(1))

            def forward(self, x):
                out = (x[:, :1] * self.w * float("inf")) * 0.0  # NaN
                return out.expand(x.shape[0], 4) if out.shape[1] == 1 else out

        m = _M()
        return Subnetwork(module=m, complexity=1.0)


class _GoodBuilder(Builder):

    def __init__(self, name="good"):
        self._name = name

    @property
    def name(self):
        return self._name

    def build_subnetwork(self, features, logits_dimension, training,
                         previous_ensemble=None):
        from adanet_amd.ops.linear import HipLinear

        class _M(nn.Module):

            def __init__(self):
                super().__init__()
                self.lin = HipLinear(features.shape[1], logits_dimension)
                self.last_layer_dim = features.shape[1]

            def forward(self, x):
                return x, self.lin(x)

        return Subnetwork(module=_M(), complexity=1.0)


def test_nan_candidate_loses_selection(model_dir, synthetic_classification):
    """Reference bookkeeping selection is np.nanargmin (estimator.py:
    1494-1512): a diverged (NaN) candidate LOSES and the healthy candidate
    is frozen — divergence does not abort the search."""
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(
        model_dir, input_fn,
        subnetwork_generator=SimpleGenerator(
            [_NanBuilder(), _GoodBuilder()]))
    est.train(input_fn, max_steps=10)
    arch = json.loads(
        open(os.path.join(model_dir, "architecture-0.json")).read())
    assert arch["subnetworks"][0]["builder_name"] == "good"


def test_all_nan_candidates_raise(model_dir, synthetic_classification):
    """ALL candidates NaN == np.nanargmin's all-NaN error: fail loudly
    (NanLossDuringTrainingError), never freeze a garbage model."""
    import pytest
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(
        model_dir, input_fn,
        subnetwork_generator=SimpleGenerator([_NanBuilder()]),
        # two candidates (the single-candidate path returns 0 without
        # inspecting losses, like the reference's early return)
        ensemble_strategies=None)
    import adanet_amd.ensemble as ens

    class _NanBuilder2(_NanBuilder):

        @property
        def name(self):
            return "nan2"

    est = _make_estimator(
        str(model_dir) + "_2", input_fn,
        subnetwork_generator=SimpleGenerator(
            [_NanBuilder(), _NanBuilder2()]))
    with pytest.raises(adanet_amd.NanLossDuringTrainingError):
        est.train(input_fn, max_steps=10)
    assert not os.path.exists(
        os.path.join(str(model_dir) + "_2", "architecture-0.json"))


def test_evaluator_selects_on_eval_loss(model_dir, synthetic_classification):
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(
        model_dir, input_fn,
        evaluator=adanet_amd.Evaluator(input_fn=input_fn, steps=3))
    est.train(input_fn, max_steps=10)
    assert os.path.exists(os.path.join(model_dir, "architecture-0.json"))


def test_report_materialization(model_dir, synthetic_classification):
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(
        model_dir, input_fn,
        report_materializer=adanet_amd.ReportMaterializer(input_fn=input_fn,
                                                          steps=1))
    est.train(input_fn, max_steps=20)
    path = os.path.join(model_dir, "report", "iteration_reports.json")
    assert os.path.exists(path)
    reports = est._report_accessor.read_iteration_reports()
    assert len(reports) == 2
    names = {r.name for r in reports[0]}
    assert "linear" in names or "1_layer_dnn" in names
    assert any(r.included_in_final_ensemble for r in reports[0])


def test_summaries_written(model_dir, synthetic_classification):
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(model_dir, input_fn)
    est.train(input_fn, max_steps=10)
    assert os.path.isdir(os.path.join(model_dir, "summaries"))


def test_multi_head_lifecycle(model_dir):
    """Multi-head estimator (reference estimator_test.py:1517)."""
    from adanet_amd.head import MultiClassHead, MultiHead

    torch.manual_seed(0)
    N, D = 256, 16
    X = torch.randn(N, D)
    W1, W2 = torch.randn(D, 3), torch.randn(D, 4)
    Y = {"a": (X @ W1).argmax(1), "b": (X @ W2).argmax(1)}

    def input_fn():
        def gen():
            g = torch.Generator().manual_seed(5)
            while True:
                idx = torch.randint(0, N, (64,), generator=g)
                yield X[idx], {k: v[idx] for k, v in Y.items()}

        return gen()

    head = MultiHead({"a": MultiClassHead(3), "b": MultiClassHead(4)})
    est = adanet_amd.Estimator(
        head=head,
        subnetwork_generator=simple_dnn.Generator(layer_size=16),
        max_iteration_steps=10, model_dir=model_dir,
        config=adanet_amd.RunConfig(tf_random_seed=3))
    est.train(input_fn, max_steps=20)
    res = est.evaluate(input_fn, steps=4)
    assert "a/accuracy" in res and "b/accuracy" in res


def test_train_hooks_and_stop(model_dir, synthetic_classification):
    """Hook callbacks + StopAfterSteps early stop (reference
    _StopAfterTrainingHook, estimator.py:50-87) and TrainOpSpec hooks."""
    from adanet_amd import hooks as hooks_lib
    from adanet_amd.subnetwork import TrainOpSpec
    X, Y, input_fn = synthetic_classification
    calls = {"begin": 0, "before": 0, "after": 0, "end": 0}

    class _H(hooks_lib.TrainHook):

        def begin(self, estimator=None, iteration=None):
            calls["begin"] += 1

        def before_step(self, gs):
            calls["before"] += 1

        def after_step(self, gs):
            calls["after"] += 1

        def end(self, estimator=None):
            calls["end"] += 1

    est = _make_estimator(model_dir, input_fn)
    stopper = hooks_lib.StopAfterSteps(4)
    est.train(input_fn, max_steps=30, hooks=[_H(), stopper])
    assert est.global_step == 4  # stopped mid-iteration 0
    assert calls["begin"] == 1 and calls["end"] == 1
    assert calls["before"] == 4 and calls["after"] == 4

    # TrainOpSpec-attached hooks reach the loop too.
    spec_calls = []

    class _SpecHook(hooks_lib.TrainHook):

        def after_step(self, gs):
            spec_calls.append(gs)

    class _HookedBuilder(_GoodBuilder):

        def build_optimizer(self, params, iteration=0):
            from adanet_amd.ops.optim import FusedSGD
            return TrainOpSpec(FusedSGD(params, lr=0.01),
                               hooks=(_SpecHook(),))

    est2 = _make_estimator(str(model_dir) + "2", input_fn,
                           subnetwork_generator=SimpleGenerator(
                               [_HookedBuilder()]))
    est2.train(input_fn, steps=3)
    assert len(spec_calls) == 3


def test_frozen_logit_cache_correctness(model_dir, synthetic_classification):
    """Cached frozen logits must give the same losses as recomputation:
    run two iterations with cache keys vs without — identical winners and
    eval losses."""
    X, Y, input_fn = synthetic_classification

    def keyed_input_fn():
        def gen():
            g = torch.Generator().manual_seed(7)
            batches = []
            for i in range(8):
                idx = torch.randint(0, X.shape[0], (64,), generator=g)
                xb = X[idx].clone()
                xb.adanet_cache_key = ("t", i)
                batches.append((xb, Y[idx]))
            i = 0
            while True:
                yield batches[i % 8]
                i += 1

        return gen()

    est1 = _make_estimator(str(model_dir) + "_keyed", keyed_input_fn)
    est1.train(keyed_input_fn, max_steps=20)
    r1 = est1.evaluate(keyed_input_fn, steps=4)

    def unkeyed_input_fn():
        def gen():
            for f, l in keyed_input_fn():
                f2 = f.clone()  # drops the cache key attribute
                yield f2, l

        return gen()

    est2 = _make_estimator(str(model_dir) + "_unkeyed", unkeyed_input_fn)
    est2.train(unkeyed_input_fn, max_steps=20)
    r2 = est2.evaluate(unkeyed_input_fn, steps=4)
    assert r1["loss"] == pytest.approx(r2["loss"], abs=1e-4)
    assert (r1["architecture/adanet/ensembles"] ==
            r2["architecture/adanet/ensembles"])


def test_frozen_logit_cache_is_cross_iteration(model_dir,
                                               synthetic_classification):
    """A frozen member's forward runs ONCE per resident batch over the
    WHOLE run, not once per iteration: the cache outlives the iteration
    (per-iteration cost stays O(new members) as the ensemble grows)."""
    X, Y, input_fn = synthetic_classification
    forward_counts = {}

    class _CountingBuilder(_GoodBuilder):

        def build_subnetwork(self, features, logits_dimension, training,
                             previous_ensemble=None):
            sub = super().build_subnetwork(features, logits_dimension,
                                           training, previous_ensemble)
            mod = sub.module
            orig_forward = mod.forward
            bname = self.name

            def counted(x, _orig=orig_forward, _b=bname):
                if not mod.training and not torch.is_grad_enabled():
                    forward_counts[_b] = forward_counts.get(_b, 0) + 1
                return _orig(x)

            mod.forward = counted
            return sub

    n_batches = 4

    def keyed_input_fn():
        def gen():
            g = torch.Generator().manual_seed(3)
            batches = []
            for i in range(n_batches):
                idx = torch.randint(0, X.shape[0], (32,), generator=g)
                xb = X[idx].clone()
                xb.adanet_cache_key = ("kb", i)
                batches.append((xb, Y[idx]))
            i = 0
            while True:
                yield batches[i % n_batches]
                i += 1

        return gen()

    est = _make_estimator(
        model_dir, keyed_input_fn,
        subnetwork_generator=SimpleGenerator([_CountingBuilder("g0")]),
        max_iteration_steps=n_batches, force_grow=True)
    baseline = None
    # 4 iterations: the member frozen at iteration 0 must NOT be re-run
    # at iterations 2 and 3 (its logits replay from the HBM cache).
    for it in range(4):
        forward_counts.clear()
        est.train(keyed_input_fn, steps=n_batches)
        if it == 1:
            # first iteration with a frozen member: populates the cache
            baseline = dict(forward_counts)
            assert baseline, "expected frozen-member eval forwards"
        if it >= 2:
            # all resident batches already cached for old members; only
            # the NEWLY frozen winner of iteration it-1 misses.
            total = sum(forward_counts.values())
            assert total <= sum(baseline.values()), (
                "frozen forwards grew with ensemble size: %r (baseline %r)"
                % (forward_counts, baseline))


def test_builder_prune_previous_ensemble(model_dir,
                                         synthetic_classification):
    """Legacy pruning hook (reference ensemble_builder.py:371-395): a
    builder that keeps no previous members yields a 1-member architecture
    even at iteration 1."""
    X, Y, input_fn = synthetic_classification

    class _PruningBuilder(_GoodBuilder):

        def prune_previous_ensemble(self, previous_ensemble):
            return []  # drop everything

    est = _make_estimator(
        model_dir, input_fn, force_grow=True,
        subnetwork_generator=SimpleGenerator([_PruningBuilder("pruner")]))
    est.train(input_fn, max_steps=20)
    arch = json.loads(
        open(os.path.join(model_dir, "architecture-1.json")).read())
    assert len(arch["subnetworks"]) == 1
    assert arch["subnetworks"][0]["iteration_number"] == 1


@pytest.mark.parametrize("mixture_type", ["scalar", "vector", "matrix"])
def test_mixture_types_end_to_end(model_dir, synthetic_classification,
                                  mixture_type):
    """Full lifecycle per MixtureWeightType incl. mixture-weight training
    (reference weighted_test.py parameter grid :85-483)."""
    import functools

    from adanet_amd.ensemble import ComplexityRegularizedEnsembler
    from adanet_amd.ops.optim import FusedSGD
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(
        model_dir, input_fn,
        ensemblers=[
            ComplexityRegularizedEnsembler(
                optimizer=functools.partial(FusedSGD, lr=0.05),
                mixture_weight_type=mixture_type,
                warm_start_mixture_weights=True,
                model_dir=model_dir,
                adanet_lambda=1e-4,
                use_bias=True)
        ])
    est.train(input_fn, max_steps=20)
    assert est.iteration_number == 2
    res = est.evaluate(input_fn, steps=4)
    assert res["loss"] == res["loss"]


def test_mean_ensembler_end_to_end(model_dir, synthetic_classification):
    from adanet_amd.ensemble import MeanEnsembler
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(model_dir, input_fn,
                          ensemblers=[MeanEnsembler()])
    est.train(input_fn, max_steps=20)
    res = est.evaluate(input_fn, steps=4)
    assert "accuracy" in res


def test_multiple_ensemblers_and_strategies(model_dir,
                                            synthetic_classification):
    """candidates x ensemblers grid (reference iteration.py:683-740) with
    Solo+Grow strategies and ComplexityRegularized+Mean ensemblers."""
    from adanet_amd.ensemble import (AllStrategy, ComplexityRegularizedEnsembler,
                                     MeanEnsembler, SoloStrategy)
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(
        model_dir, input_fn,
        ensemblers=[ComplexityRegularizedEnsembler(), MeanEnsembler()],
        ensemble_strategies=[SoloStrategy(), AllStrategy()])
    est.train(input_fn, max_steps=10)
    it = est._current_iteration
    # strategies: solo -> 2 candidates, all -> 1; x2 ensemblers = 6 specs
    # (iteration 0: no previous-best candidate).
    assert est.iteration_number == 1
    arch = json.loads(
        open(os.path.join(model_dir, "architecture-0.json")).read())
    assert arch["ensembler_name"] in ("complexity_regularized", "mean")


def test_duplicate_ensembler_names_raise(model_dir):
    from adanet_amd.ensemble import MeanEnsembler
    with pytest.raises(ValueError):
        _make_estimator(model_dir, None,
                        ensemblers=[MeanEnsembler(), MeanEnsembler()])


def test_regression_head_end_to_end(model_dir):
    from adanet_amd.head import RegressionHead
    torch.manual_seed(0)
    N, D = 256, 8
    X = torch.randn(N, D)
    Y = X @ torch.randn(D)

    def input_fn():
        def gen():
            g = torch.Generator().manual_seed(5)
            while True:
                idx = torch.randint(0, N, (64,), generator=g)
                yield X[idx], Y[idx]

        return gen()

    est = _make_estimator(model_dir, input_fn, head=RegressionHead())
    est.train(input_fn, max_steps=20)
    res = est.evaluate(input_fn, steps=4)
    assert res["average_loss"] == res["average_loss"]


def test_keep_checkpoint_max_gc(model_dir, synthetic_classification):
    """Old increment checkpoints are garbage-collected down to
    keep_checkpoint_max (reference keep_checkpoint_max semantics)."""
    X, Y, input_fn = synthetic_classification
    gen = simple_dnn.Generator(layer_size=8)
    est = adanet_amd.Estimator(
        head=MultiClassHead(4), subnetwork_generator=gen,
        max_iteration_steps=5, model_dir=model_dir,
        config=adanet_amd.RunConfig(tf_random_seed=1, keep_checkpoint_max=2))
    est.train(input_fn, max_steps=25)  # 5 iterations
    ckpts = glob.glob(os.path.join(model_dir, "increment.ckpt-*.pt"))
    assert len(ckpts) == 2, sorted(ckpts)
    # latest checkpoint pointer still resolves
    with open(os.path.join(model_dir, "checkpoint")) as f:
        assert os.path.exists(f.read().strip())


def test_resume_across_processes(model_dir, tmp_path):
    """Kill-and-restart robustness: a FRESH PROCESS resumes mid-iteration
    from disk alone (no in-memory state survives)."""
    import subprocess
    import sys
    script = tmp_path / "resume_driver.py"
    script.write_text("""
import sys
import torch
import adanet_amd
from adanet_amd.head import MultiClassHead
from adanet_amd.models import simple_dnn

model_dir, steps = sys.argv[1], int(sys.argv[2])
torch.manual_seed(0)
X = torch.randn(64, 12)
Y = (X[:, :4].sum(1) > 0).long() + 2 * (X[:, 4:8].sum(1) > 0).long()

def input_fn():
    while True:
        yield X, Y

est = adanet_amd.Estimator(
    head=MultiClassHead(4),
    subnetwork_generator=simple_dnn.Generator(layer_size=8),
    max_iteration_steps=10, model_dir=model_dir,
    config=adanet_amd.RunConfig(tf_random_seed=7))
est.train(input_fn, steps=steps)
print("ITER=%d STEP=%d" % (est.iteration_number, est.global_step))
""")
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

    def run(steps):
        env = dict(os.environ)
        env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
        out = subprocess.run(
            [sys.executable, str(script), model_dir, str(steps)],
            capture_output=True, text=True, timeout=600, env=env)
        assert out.returncode == 0, out.stderr[-2000:]
        return out.stdout.strip().splitlines()[-1]

    assert run(12) == "ITER=1 STEP=12"   # process A dies mid-iteration 1
    assert run(18) == "ITER=3 STEP=30"   # process B resumes + continues


def test_adanet_lambda_steers_selection_toward_low_complexity(
        model_dir, synthetic_classification):
    """The complexity penalty changes WHICH architectures win: with a
    strong lambda, zero-complexity (linear) candidates beat deeper DNNs
    (the reference adanet_objective tutorial's headline behavior)."""
    X, Y, input_fn = synthetic_classification

    def total_complexity(lam, sub_dir):
        est = adanet_amd.Estimator(
            head=MultiClassHead(4),
            subnetwork_generator=simple_dnn.Generator(
                layer_size=8, learn_mixture_weights=True),
            max_iteration_steps=15, adanet_lambda=lam,
            model_dir=os.path.join(model_dir, sub_dir),
            config=adanet_amd.RunConfig(tf_random_seed=11))
        est.train(input_fn, max_steps=45)
        ens, _ = est._rebuild_previous_ensemble(est.iteration_number, X[:8])
        return sum(float(ws.subnetwork.complexity)
                   for ws in ens.weighted_subnetworks)

    assert total_complexity(5.0, "strong") <= total_complexity(0.0, "none")


def test_replicate_ensemble_in_training(model_dir, synthetic_classification):
    """replicate_ensemble_in_training=True recomputes frozen members in
    TRAIN mode each step (reference iteration.py:569-572 semantics: no
    eval-mode freeze); training still completes and grows."""
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(model_dir, input_fn,
                          replicate_ensemble_in_training=True)
    est.train(input_fn, max_steps=20)
    assert est.iteration_number == 2
    res = est.evaluate(input_fn, steps=3)
    assert math.isfinite(res["loss"])


def test_save_checkpoints_steps_mid_iteration(model_dir,
                                              synthetic_classification):
    """save_checkpoints_steps writes MID-iteration checkpoints (resume
    granularity below one boosting round)."""
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(
        model_dir, input_fn,
        config=adanet_amd.RunConfig(tf_random_seed=42,
                                    save_checkpoints_steps=4))
    est.train(input_fn, steps=6)  # inside iteration 0
    est._join_ckpt_writer()
    assert glob.glob(os.path.join(model_dir, "increment.ckpt-0.pt"))
    est2 = _make_estimator(model_dir, input_fn)
    assert est2.global_step in (4, 6)  # resumed from a mid-iteration save


def test_seeded_runs_reproduce_architectures(model_dir,
                                             synthetic_classification):
    """Same tf_random_seed -> identical selected architectures across two
    independent runs (CPU is deterministic; the reference's determinism
    contract for Generators, generator.py:288)."""
    X, Y, input_fn = synthetic_classification

    def run(sub):
        d = os.path.join(model_dir, sub)
        est = _make_estimator(d, input_fn)
        est.train(input_fn, max_steps=30)
        return [json.loads(open(os.path.join(
            d, "architecture-%d.json" % t)).read())["subnetworks"]
            for t in range(3)]

    assert run("a") == run("b")


def test_custom_metric_fn(model_dir, synthetic_classification):
    """metric_fn(predictions, features, labels) merges into evaluate()
    results (reference estimator metric_fn contract)."""
    X, Y, input_fn = synthetic_classification

    def metric_fn(predictions, features, labels):
        top1 = predictions["class_ids"]
        return {"custom_error": float((top1 != labels).float().mean())}

    est = _make_estimator(model_dir, input_fn, metric_fn=metric_fn)
    est.train(input_fn, max_steps=20)
    res = est.evaluate(input_fn, steps=3)
    assert "custom_error" in res
    assert res["custom_error"] == pytest.approx(1.0 - res["accuracy"],
                                                abs=1e-6)


def test_evaluator_custom_metric_maximize(model_dir,
                                          synthetic_classification):
    """Evaluator(metric_name='accuracy', objective=MAXIMIZE) drives
    selection by head accuracy instead of adanet_loss."""
    X, Y, input_fn = synthetic_classification
    est = _make_estimator(
        model_dir, input_fn,
        evaluator=adanet_amd.Evaluator(input_fn=input_fn, steps=2,
                                       metric_name="accuracy",
                                       objective="maximize"))
    est.train(input_fn, max_steps=20)
    assert est.iteration_number == 2
    # unknown metric raises loudly
    est2 = _make_estimator(
        os.path.join(model_dir, "bad"), input_fn,
        evaluator=adanet_amd.Evaluator(input_fn=input_fn, steps=1,
                                       metric_name="not_a_metric"))
    with pytest.raises(ValueError):
        est2.train(input_fn, max_steps=10)


def test_input_exhaustion_ends_training(model_dir, synthetic_classification):
    """max_iteration_steps=None + finite input: training consumes the
    input and returns mid-iteration (reference OutOfRange semantics —
    no bookkeeping runs for the incomplete iteration)."""
    X, Y, input_fn = synthetic_classification

    def finite_input_fn():
        return iter([(X[:32], Y[:32])] * 7)

    est = _make_estimator(model_dir, finite_input_fn,
                          max_iteration_steps=None)
    est.train(finite_input_fn)
    assert est.global_step == 7
    assert est.iteration_number == 0
    # resume with more data completes nothing new until budget exists
    est.train(finite_input_fn)
    assert est.global_step == 14


def test_debug_mode_raises_on_nonfinite_input(model_dir):
    """debug=True validates every input batch (reference debug mode,
    estimator_test.py:3081 family)."""
    X = torch.randn(32, 8)
    X[3, 4] = float("inf")
    Y = torch.zeros(32, dtype=torch.long)

    def input_fn():
        def gen():
            while True:
                yield X, Y
        return gen()

    est = _make_estimator(model_dir, input_fn, debug=True)
    with pytest.raises(ValueError):
        est.train(input_fn, max_steps=5)


@pytest.mark.parametrize("force_grow,use_replay,use_evaluator", [
    (True, False, False),
    (True, False, True),
    (False, False, False),
    (True, True, False),   # replay overrides force_grow (reference
    (True, True, True),    # early-return, estimator.py:1433-1438)
    (False, True, False),
])
def test_selection_contract_grid(model_dir, synthetic_classification,
                                 force_grow, use_replay, use_evaluator):
    """Reference golden selection behaviors (estimator.py:1415-1517) over
    {force_grow x replay x evaluator-absent}:
      * force_grow: the previous ensemble never wins when growth is
        possible -> member count strictly grows each iteration;
      * replay pins the winner index and is NEVER re-routed by force_grow
        (a replayed index 0 at t>0 keeps the previous ensemble even with
        force_grow=True);
      * evaluator presence only changes the loss source, not the rules."""
    X, Y, input_fn = synthetic_classification
    kwargs = dict(
        subnetwork_generator=SimpleGenerator(
            [_GoodBuilder("g0"), _GoodBuilder("g1")]),
        force_grow=force_grow,
    )
    if use_replay:
        import adanet_amd.replay as replay
        kwargs["replay_config"] = replay.Config(
            best_ensemble_indices=[0, 0, 0])
    if use_evaluator:
        kwargs["evaluator"] = adanet_amd.Evaluator(input_fn=input_fn,
                                                   steps=2)
    est = _make_estimator(model_dir, input_fn, **kwargs)
    est.train(input_fn, max_steps=30)  # 3 iterations x 10 steps
    sizes = []
    for t in range(3):
        arch = json.loads(open(os.path.join(
            model_dir, "architecture-%d.json" % t)).read())
        sizes.append(len(arch["subnetworks"]))
    if use_replay:
        # replayed index 0: t0 picks candidate g0 (1 member); t>=1 index 0
        # is the previous ensemble -> the ensemble NEVER grows, even with
        # force_grow on.
        assert sizes == [1, 1, 1], sizes
    elif force_grow:
        # growth every iteration, previous ensemble never kept
        assert sizes == [1, 2, 3], sizes
    else:
        # monotone non-shrinking; may plateau when previous wins
        assert all(b >= a for a, b in zip(sizes, sizes[1:])), sizes


def test_best_metrics_mux_with_dict_metrics(model_dir,
                                            synthetic_classification):
    """Dict-valued metric_fn flows through the per-candidate stores and the
    best-mux returns the WINNER's dict (reference eval_metrics.py:306-408
    best_eval_metrics_tuple semantics)."""
    X, Y, input_fn = synthetic_classification

    def metric_fn(predictions, features, labels):
        probs = predictions["probabilities"]
        return {
            "custom/max_prob": float(probs.max()),
            "custom/n_examples": float(probs.shape[0]),
        }

    est = _make_estimator(model_dir, input_fn, metric_fn=metric_fn)
    est.train(input_fn, max_steps=10)
    res = est.evaluate(input_fn, steps=2)
    assert "custom/max_prob" in res and 0 < res["custom/max_prob"] <= 1.0
    assert res["custom/n_examples"] > 0
    assert "best_ensemble_index_0" in res
    assert "architecture/adanet/ensembles" in res


def test_golden_deterministic_lifecycle(model_dir, tmp_path,
                                        synthetic_classification):
    """Bit-exact repeatability of a full two-iteration search on CPU
    (the reference pins golden losses per config, estimator_test.py
    :419-470; with the round-2 deterministic reductions the equivalent
    guarantee here is byte-equality of the entire trajectory). Two
    estimators with the same seed must produce identical architectures
    AND identical evaluate() numbers."""
    import torch
    X, Y, input_fn = synthetic_classification
    n_threads = torch.get_num_threads()
    torch.set_num_threads(1)  # CPU reductions vary with thread count
    try:
        results = []
        for d in (model_dir, str(tmp_path / "golden_b")):
            est = _make_estimator(d, input_fn)
            est.train(input_fn, max_steps=20)
            arch = json.load(open(os.path.join(d, "architecture-1.json")))
            res = est.evaluate(input_fn, steps=2)
            results.append((
                [s["builder_name"] for s in arch["subnetworks"]],
                float(res["loss"]), float(res.get("accuracy", -1.0))))
        assert results[0][0] == results[1][0]
        assert results[0][1] == results[1][1]  # bit-exact, not approx
        assert results[0][2] == results[1][2]
    finally:
        torch.set_num_threads(n_threads)
