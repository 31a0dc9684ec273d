"""Aux-subsystem unit tests mirroring the reference's per-module suites:
report accessor JSON round-trips incl. unicode (report_accessor_test.py:248),
scoped summary event isolation (summary_test.py pattern), countdown timer."""

import json
import os
import time

import pytest

from adanet_amd.core.report_accessor import _ReportAccessor
from adanet_amd.core.summary import _ScopedSummary
from adanet_amd.core.timer import _CountDownTimer
from adanet_amd.subnetwork.report import MaterializedReport


def test_report_accessor_round_trip_and_unicode(tmp_path):
    acc = _ReportAccessor(str(tmp_path / "report"))
    r0 = MaterializedReport(
        iteration_number=0, name="dnn_é中文",
        hparams={"lr": 0.1, "layers": 2},
        attributes={"note": "✓ included"},
        metrics={"loss": 1.5}, included_in_final_ensemble=True)
    acc.write_iteration_report(0, [r0])
    acc.write_iteration_report(1, [
        MaterializedReport(iteration_number=1, name="b", metrics={"loss": 1.0})
    ])
    got = acc.read_iteration_reports()
    assert len(got) == 2
    assert got[0][0].name == r0.name  # unicode survives
    assert got[0][0].attributes["note"] == "✓ included"
    assert got[0][0].included_in_final_ensemble is True
    # idempotent overwrite per iteration (reference overwrite-on-retrain)
    acc.write_iteration_report(1, [
        MaterializedReport(iteration_number=1, name="b2",
                           metrics={"loss": 0.9})
    ])
    got = acc.read_iteration_reports()
    assert len(got) == 2 and got[1][0].name == "b2"
    # file is plain JSON, iteration-keyed
    raw = json.loads(open(acc.report_file_path).read())
    assert set(raw) == {"0", "1"}


def test_scoped_summary_isolation_and_family(tmp_path):
    s_a = _ScopedSummary(str(tmp_path), scope="cand_a")
    s_b = _ScopedSummary(str(tmp_path), scope="cand_b")
    s_a.set_step(3)
    s_b.set_step(3)
    s_a.scalar("loss", 1.25)
    s_b.scalar("loss", 2.5)
    s_a.scalar("adanet_loss", 0.5, family="eval")
    events_a, events_b = [], []
    for scope, sink in [("cand_a", events_a), ("cand_b", events_b)]:
        path = os.path.join(str(tmp_path), scope, "events.jsonl")
        assert os.path.exists(path), path
        with open(path) as f:
            sink.extend(json.loads(l) for l in f if l.strip())
    tags_a = {e["tag"]: e for e in events_a}
    assert tags_a["loss"]["value"] == 1.25 and tags_a["loss"]["step"] == 3
    assert "eval/adanet_loss" in tags_a  # family prefixes the tag
    assert {e["tag"] for e in events_b} == {"loss"}
    assert [e["value"] for e in events_b] == [2.5]


def test_countdown_timer():
    t = _CountDownTimer(0.2)
    assert t.secs_remaining() > 0
    time.sleep(0.25)
    assert t.secs_remaining() == 0


def test_public_api_surface():
    """Every symbol the docs/README reference resolves (reference
    adanet_test.py:23-25 pattern: assert public attrs exist)."""
    import adanet_amd
    from adanet_amd import distributed, ensemble, replay, serving, subnetwork
    for sym in ("Estimator", "AutoEnsembleEstimator",
                "AutoEnsembleSubestimator", "Evaluator",
                "ReportMaterializer", "RunConfig", "Summary"):
        assert hasattr(adanet_amd, sym), sym
    assert distributed.RoundRobinStrategy and distributed.ReplicationStrategy
    assert serving.export_torchscript and serving.export_program
    assert replay.Config and subnetwork.Builder and subnetwork.Generator
    assert ensemble.ComplexityRegularizedEnsembler and ensemble.MeanEnsembler
    from adanet_amd.experimental import (ModelSearch,  # noqa: F401
                                         MultiGpuScheduler,
                                         ThreadedScheduler)
    from adanet_amd.head import (BinaryClassHead,  # noqa: F401
                                 MultiClassHead, MultiHead, RegressionHead)
    from adanet_amd.ops.conv import (HipConv1x1,  # noqa: F401
                                     HipConvNxN, HipDepthwiseConv2d)


def test_runconfig_env_parsing(monkeypatch):
    from adanet_amd.config import RunConfig
    monkeypatch.setenv("WORLD_SIZE", "8")
    monkeypatch.setenv("RANK", "3")
    c = RunConfig(tf_random_seed=5)
    assert c.world_size == 8 and c.rank == 3
    assert c.random_seed == 5  # reference kwarg name preserved
    monkeypatch.delenv("WORLD_SIZE")
    monkeypatch.delenv("RANK")
    c2 = RunConfig()
    assert c2.world_size == 1 and c2.rank == 0


def test_train_manager_persistence(tmp_path):
    """TrainManager round-trips 'done' state through its JSON dir
    (reference iteration.py:40-118 restart semantics)."""
    from adanet_amd.core.iteration import _TrainManager
    tm = _TrainManager(str(tmp_path), 2, is_chief=True)
    assert tm.should_train("cand_a")
    tm.request_stop("cand_a", "NaN loss")
    tm2 = _TrainManager(str(tmp_path), 2, is_chief=True)  # fresh process
    assert not tm2.should_train("cand_a")
    assert tm2.stopped["cand_a"] == "NaN loss"
    assert tm2.should_train("cand_b")
    assert not tm2.is_over(["cand_a", "cand_b"])
    tm2.request_stop("cand_b")
    assert tm2.is_over(["cand_a", "cand_b"])
