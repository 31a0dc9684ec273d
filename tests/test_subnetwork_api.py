"""Search-space API validation tables (reference test model:
adanet/subnetwork/generator_test.py:37-220, report_test.py:32-125)."""

import pytest
import torch
from torch import nn

from adanet_amd.subnetwork import (Builder, Report, SimpleGenerator,
                                   Subnetwork)


class _Mod(nn.Module):

    def forward(self, x):
        return x, x


class _B(Builder):

    def __init__(self, name="b"):
        self._name = name

    @property
    def name(self):
        return self._name

    def build_subnetwork(self, features, logits_dimension, training,
                         previous_ensemble=None):
        return Subnetwork(module=_Mod(), complexity=1.0)


def test_subnetwork_requires_module():
    with pytest.raises(ValueError):
        Subnetwork(module="not a module")


def test_subnetwork_tensor_complexity_materialized():
    s = Subnetwork(module=_Mod(), complexity=torch.tensor(2.0))
    assert s.complexity == 2.0


def test_simple_generator_empty_raises():
    with pytest.raises(ValueError):
        SimpleGenerator([])


def test_simple_generator_non_builder_raises():
    with pytest.raises(ValueError):
        SimpleGenerator(["not a builder"])


def test_simple_generator_returns_constant_list():
    builders = [_B("a"), _B("b")]
    gen = SimpleGenerator(builders)
    out = gen.generate_candidates(None, 0, [], [])
    assert [b.name for b in out] == ["a", "b"]
    out2 = gen.generate_candidates(None, 3, [], [])
    assert out == out2


@pytest.mark.parametrize("hparams,ok", [
    ({"x": 1}, True),
    ({"x": 1.5}, True),
    ({"x": True}, True),
    ({"x": "s"}, True),
    ({"x": [1]}, False),
    ({"x": None}, False),
    ({"x": torch.tensor(1.0)}, False),
])
def test_report_hparam_validation(hparams, ok):
    if ok:
        Report(hparams=hparams, attributes={}, metrics={})
    else:
        with pytest.raises(ValueError):
            Report(hparams=hparams, attributes={}, metrics={})


def test_report_attributes_allow_scalar_tensors():
    r = Report(hparams={}, attributes={"t": torch.tensor(3.0)},
               metrics={"m": torch.tensor(1)})
    m = r.materialize(2, "sub", included_in_final_ensemble=True)
    assert m.attributes["t"] == 3.0
    assert m.metrics["m"] == 1
    assert m.iteration_number == 2 and m.name == "sub"
    assert m.included_in_final_ensemble
    # JSON round-trip
    from adanet_amd.subnetwork import MaterializedReport
    assert MaterializedReport.from_json(m.to_json()) == m


def test_report_rejects_nonscalar_tensor():
    with pytest.raises(ValueError):
        Report(hparams={}, attributes={"t": torch.ones(3)}, metrics={})


def test_builder_optional_summary_param(tmp_path):
    """Builders declaring a `summary` param receive the candidate's scoped
    summary during training builds (reference Builder signature,
    generator.py:162-270) and a disabled one on eval-mode rebuilds."""
    import adanet_amd
    from adanet_amd.head import MultiClassHead
    from adanet_amd.subnetwork import Builder, SimpleGenerator, Subnetwork
    import torch
    from torch import nn

    seen = []

    class _B(Builder):

        @property
        def name(self):
            return "with_summary"

        def build_subnetwork(self, features, logits_dimension, training,
                             previous_ensemble=None, summary=None):
            seen.append(summary)
            if summary is not None:
                summary.scalar("built", 1.0)
            lin = nn.Linear(features.shape[1], logits_dimension)
            m = nn.Module()
            m.lin = lin
            m.forward = lambda x, _l=lin: (x, _l(x))
            return Subnetwork(module=m, complexity=1.0)

    torch.manual_seed(0)
    X = torch.randn(64, 8)
    Y = (X.sum(dim=1) > 0).long()

    def input_fn():
        def gen():
            while True:
                yield X, Y
        return gen()

    est = adanet_amd.Estimator(
        head=MultiClassHead(2), subnetwork_generator=SimpleGenerator([_B()]),
        max_iteration_steps=5, model_dir=str(tmp_path / "m"),
        config=adanet_amd.RunConfig(tf_random_seed=0))
    est.train(input_fn, max_steps=10)
    assert seen and all(s is not None for s in seen)
