"""Unit tests: architecture JSON, report accessor, evaluator, candidate EMA,
summaries, timer (reference test model: architecture_test.py:27-82,
report_accessor_test.py:28-297, evaluator_test.py:51-133, candidate_test.py,
timer_test.py, summary_test.py)."""

import json
import math
import os
import time

import pytest
import torch

from adanet_amd.core.architecture import _Architecture
from adanet_amd.core.candidate import _Candidate
from adanet_amd.core.evaluator import Evaluator, Objective
from adanet_amd.core.report_accessor import _ReportAccessor
from adanet_amd.core.summary import _ScopedSummary, read_events
from adanet_amd.core.timer import _CountDownTimer
from adanet_amd.subnetwork.report import MaterializedReport


def test_architecture_roundtrip():
    a = _Architecture("cand", "complexity_regularized")
    a.add_subnetwork(0, "linear")
    a.add_subnetwork(1, "dnn")
    a.add_subnetwork(1, "wide")
    a.set_replay_indices([0, 2])
    s = a.serialize(iteration_number=2, global_step=300)
    parsed = json.loads(s)
    assert parsed["iteration_number"] == 2
    assert parsed["global_step"] == 300
    assert parsed["ensemble_candidate_name"] == "cand"
    b = _Architecture.deserialize(s)
    assert b.subnetworks == ((0, "linear"), (1, "dnn"), (1, "wide"))
    assert b.replay_indices == [0, 2]
    assert b.subnetworks_grouped_by_iteration == ((0, ("linear",)),
                                                  (1, ("dnn", "wide")))


def test_architecture_serialize_is_sorted_json():
    a = _Architecture("c", "e")
    s = a.serialize(0, 0)
    assert s == json.dumps(json.loads(s), sort_keys=True)


def test_report_accessor_roundtrip(tmp_path):
    acc = _ReportAccessor(str(tmp_path / "report"))
    r0 = MaterializedReport(0, "a", {"h": 1}, {"at": "x"}, {"m": 0.5}, True)
    r1 = MaterializedReport(0, "bé", {}, {}, {}, False)  # unicode
    acc.write_iteration_report(0, [r0, r1])
    acc.write_iteration_report(1, [MaterializedReport(1, "c", {}, {}, {})])
    got = acc.read_iteration_reports()
    assert len(got) == 2
    assert got[0][0] == r0
    assert got[0][1].name == "bé"
    # Overwrite iteration 1 (idempotent rewrite)
    acc.write_iteration_report(1, [MaterializedReport(1, "d", {}, {}, {})])
    got = acc.read_iteration_reports()
    assert [r.name for r in got[1]] == ["d"]


def test_evaluator_minimize_maximize():
    data = [(torch.zeros(2, 2), torch.zeros(2)) for _ in range(3)]

    def input_fn():
        return iter(list(data))

    ev = Evaluator(input_fn=input_fn, steps=3)
    vals = ev.evaluate([lambda f, l: 1.0, lambda f, l: 0.5])
    assert vals == [1.0, 0.5]
    assert ev.best_index(vals) == 1
    ev2 = Evaluator(input_fn=input_fn, steps=3,
                    objective=Objective.MAXIMIZE)
    assert ev2.best_index(vals) == 0
    with pytest.raises(ValueError):
        Evaluator(input_fn=input_fn, objective="bogus")


def test_evaluator_steps_none_runs_to_exhaustion():
    data = [(torch.zeros(1), torch.zeros(1))] * 5

    def input_fn():
        return iter(list(data))

    ev = Evaluator(input_fn=input_fn, steps=None)
    calls = []
    ev.evaluate([lambda f, l: calls.append(1) or 1.0])
    assert len(calls) == 5


def test_candidate_ema():
    class _S:
        name = "s"

    c = _Candidate(_S(), adanet_loss_decay=0.5)
    assert c.adanet_loss == float("inf")
    c.update(1.0)
    assert c.adanet_loss == 1.0  # first value seeds the EMA
    c.update(3.0)
    assert c.adanet_loss == pytest.approx(2.0)
    c.update(float("nan"))
    assert math.isnan(c.adanet_loss)
    c.update(1.0)  # NaN poisons permanently (assign_moving_average semantics)
    assert c.adanet_loss == 1.0 or math.isnan(c.adanet_loss)


def test_candidate_decay_validation():
    class _S:
        name = "s"

    with pytest.raises(ValueError):
        _Candidate(_S(), adanet_loss_decay=1.5)


def test_scoped_summary_isolation(tmp_path):
    s1 = _ScopedSummary(str(tmp_path), scope="cand1", namespace="t0_ensemble")
    s2 = _ScopedSummary(str(tmp_path), scope="cand2", namespace="t0_ensemble")
    s1.set_step(5)
    s1.scalar("loss", 0.25)
    s1.scalar("adanet_loss", torch.tensor(0.5))
    s2.scalar("loss", 0.75)
    e1 = read_events(os.path.join(str(tmp_path), "t0_ensemble", "cand1"))
    e2 = read_events(os.path.join(str(tmp_path), "t0_ensemble", "cand2"))
    assert [e["value"] for e in e1 if e["tag"] == "loss"] == [0.25]
    assert e1[0]["step"] == 5
    assert [e["value"] for e in e2 if e["tag"] == "loss"] == [0.75]
    # same relative tag names -> overlayable charts
    assert {e["tag"] for e in e1} >= {"loss", "adanet_loss"}


def test_scoped_summary_skip():
    s = _ScopedSummary(None, scope="x")
    s.scalar("loss", 1.0)  # no-op, no crash
    s.histogram("h", torch.ones(4))


def test_scoped_summary_histogram(tmp_path):
    s = _ScopedSummary(str(tmp_path), scope="c")
    s.histogram("w", torch.tensor([1.0, 2.0, 3.0]))
    ev = read_events(os.path.join(str(tmp_path), "c"))
    assert ev[0]["value"]["mean"] == pytest.approx(2.0)
    assert ev[0]["value"]["count"] == 3


def test_countdown_timer():
    t = _CountDownTimer(10.0)
    assert 9.0 < t.secs_remaining() <= 10.0
    t2 = _CountDownTimer(0.0)
    assert t2.secs_remaining() == 0.0


def test_device_loader_cpu_passthrough():
    from adanet_amd.data import DeviceLoader
    data = [(torch.ones(2, 3), torch.zeros(2))] * 3

    def input_fn():
        return iter(list(data))

    loader = DeviceLoader(input_fn, device="cpu")
    out = list(loader())
    assert len(out) == 3
    assert torch.equal(out[0][0], torch.ones(2, 3))


def test_candidate_nan_poisons_ema():
    """A NaN loss permanently poisons a candidate's EMA so selection maps
    it to -inf (reference _NanLossHook + NaN->-inf semantics)."""
    from types import SimpleNamespace
    from adanet_amd.core.candidate import _Candidate
    c = _Candidate(SimpleNamespace(name="x"), adanet_loss_decay=0.9)
    c.update(1.0)
    c.update(float("nan"))
    c.update(0.5)  # cannot recover
    import math
    assert math.isnan(c.adanet_loss)


def test_mean_accumulator_weighted():
    from adanet_amd.core.eval_metrics import _MeanAccumulator
    a = _MeanAccumulator()
    a.update(1.0, n=3)
    a.update(5.0, n=1)
    assert a.value == 2.0


def test_merge_candidate_losses_maximize_skips_placeholders():
    """MAXIMIZE must not negate the +inf placeholders of candidates no rank
    evaluated — an unbuilt candidate could otherwise win argmin as -inf."""
    from adanet_amd.core.estimator import _merge_candidate_losses
    merged = _merge_candidate_losses([{0: 0.9}, {2: 0.7}], 4, "maximize")
    assert merged[0] == -0.9
    assert merged[1] == float("inf")  # unevaluated: stays +inf
    assert merged[2] == -0.7
    assert merged[3] == float("inf")
    # NaN stays NaN under maximize (divergence surfaces in selection).
    merged = _merge_candidate_losses([{1: float("nan")}], 2, "maximize")
    assert math.isnan(merged[1])
    # minimize: plain merge.
    merged = _merge_candidate_losses([{0: 0.5}], 2, "minimize")
    assert merged == [0.5, float("inf")]


def test_colocation_groups_union_by_candidate():
    """Builders sharing an ensemble candidate collapse into one placement
    group so round-robin never splits a candidate across ranks."""
    from adanet_amd.core.estimator import _colocation_groups

    class _B:
        def __init__(self, name):
            self.name = name

    class _C:
        def __init__(self, builders):
            self.subnetwork_builders = builders

    b = [_B("a"), _B("b"), _B("c"), _B("d")]
    # Grow-style: each candidate has one new builder -> identity groups.
    groups = _colocation_groups(b, [_C([b[0]]), _C([b[1]]), _C([b[2]]),
                                    _C([b[3]])])
    assert sorted(set(groups.values())) == [0, 1, 2, 3]
    # All-style: one candidate spans b and c -> they share a group.
    groups = _colocation_groups(b, [_C([b[0]]), _C([b[1], b[2]]),
                                    _C([b[3]])])
    assert groups["b"] == groups["c"]
    assert len(set(groups.values())) == 3
    # Transitive union: (a,b) + (b,c) -> {a,b,c} one group.
    groups = _colocation_groups(b, [_C([b[0], b[1]]), _C([b[1], b[2]])])
    assert groups["a"] == groups["b"] == groups["c"]
    assert groups["d"] != groups["a"]


def test_tb_writer_crc32c_known_vector():
    from adanet_amd.core.tb_writer import crc32c
    # RFC 3720 Castagnoli test vector
    assert crc32c(b"123456789") == 0xE3069283
    assert crc32c(b"") == 0


def test_tb_writer_roundtrip(tmp_path):
    """tfevents framing + protobuf encode must parse back CRC-clean: first
    record is the brain.Event:2 version stamp, then scalars/histograms/text
    with exact tags, steps and values (reference _ScopedSummaryV2 writes
    real event files, adanet/core/summary.py:375-637)."""
    from adanet_amd.core.tb_writer import TBEventWriter, read_tfevents
    w = TBEventWriter(str(tmp_path))
    w.scalar("loss", 0.5, step=3)
    w.scalar("eval/adanet_loss", -1.25, step=10)
    w.histogram("weights", [0.0, 1.0, 2.0, 2.0], step=4)
    w.text("architecture/adanet/ensembles", '{"x":1}', step=7)
    events = read_tfevents(w.path)
    assert events[0]["file_version"] == "brain.Event:2"
    assert events[1]["values"][0]["tag"] == "loss"
    assert events[1]["values"][0]["simple_value"] == pytest.approx(0.5)
    assert events[1]["step"] == 3
    assert events[2]["values"][0]["tag"] == "eval/adanet_loss"
    assert events[2]["values"][0]["simple_value"] == pytest.approx(-1.25)
    h = events[3]["values"][0]["histo"]
    assert h["min"] == 0.0 and h["max"] == 2.0 and h["num"] == 4
    assert h["sum"] == 5.0 and h["sum_squares"] == 9.0
    assert events[4]["values"][0]["text"] == '{"x":1}'


def test_scoped_summary_writes_tfevents_per_scope(tmp_path):
    """Each candidate scope gets its OWN event dir with the SAME tag names
    so TensorBoard overlays the charts (reference summary.py:262-296)."""
    from adanet_amd.core.tb_writer import read_tfevents
    import glob as _glob
    scopes = {}
    for scope in ("cand_a", "cand_b"):
        s = _ScopedSummary(str(tmp_path), scope=scope, namespace="t0_ens")
        s.set_step(5)
        s.scalar("adanet_loss", 0.25 if scope == "cand_a" else 0.75)
        scopes[scope] = s
    for scope, want in (("cand_a", 0.25), ("cand_b", 0.75)):
        files = _glob.glob(
            os.path.join(str(tmp_path), "t0_ens", scope,
                         "events.out.tfevents.*"))
        assert len(files) == 1, files
        events = read_tfevents(files[0])
        vals = [v for e in events for v in e["values"]]
        assert vals[0]["tag"] == "adanet_loss"
        assert vals[0]["simple_value"] == pytest.approx(want)
        assert events[1]["step"] == 5
    # the JSONL stream stays (both backends active)
    assert read_events(
        os.path.join(str(tmp_path), "t0_ens", "cand_a"))
