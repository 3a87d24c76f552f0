"""Coverage for less-traveled surfaces: exotic policy layers, the
ModifyingRestart meta-searcher, ObsNormLayer, and the import-guarded
external logger sinks."""

import math
import sys
import types

import pytest
import torch

from evotorch_amd import Problem
from evotorch_amd.decorators import vectorized


@vectorized
def sphere(x):
    return (x**2).sum(-1)


def test_structured_control_net_forward_and_dsl():
    from evotorch_amd.models import StructuredControlNet, count_parameters, str_to_net

    net = StructuredControlNet(in_features=6, out_features=3, num_layers=2, hidden_size=8)
    y = net(torch.randn(6))
    assert y.shape == (3,)
    yb = net(torch.randn(5, 6))
    assert yb.shape == (5, 3)
    # linear stream + nonlinear stream sum: zeroing the MLP leaves the linear part
    with torch.no_grad():
        for p in net._mlp.parameters():
            p.zero_()
    x = torch.randn(6)
    torch.testing.assert_close(net(x), net._linear(x))
    dsl = str_to_net("StructuredControlNet(in_features=4, out_features=2, num_layers=1, hidden_size=5)")
    assert dsl(torch.randn(4)).shape == (2,)
    assert count_parameters(dsl) > 0


def test_locomotor_net_forward_advances_time():
    from evotorch_amd.models import LocomotorNet

    net = LocomotorNet(in_features=4, out_features=2, num_sinusoids=3)
    y0 = net(torch.randn(4))
    assert y0.shape == (2,)
    x = torch.zeros(4)
    a = net(x)
    b = net(x)  # time advanced -> sinusoid stream moved
    assert not torch.allclose(a, b)


def test_obs_norm_layer_matches_running_norm():
    from evotorch_amd.neuroevolution import RunningNorm

    rn = RunningNorm(shape=5)
    rn.update(torch.randn(100, 5) * 3 + 1)
    layer = rn.to_layer()
    x = torch.randn(7, 5)
    torch.testing.assert_close(layer(x), rn.normalize(x).to(torch.float32))


def test_modifying_restart_applies_modification():
    from evotorch_amd.algorithms import CEM, ModifyingRestart

    prob = Problem("min", sphere, solution_length=5, initial_bounds=(-1, 1), seed=3)
    seen = []

    def modify(r):
        seen.append(r.num_restarts)
        r.algorithm_args["popsize"] = r.algorithm_args.get("popsize", 20) + 10

    r = ModifyingRestart(
        prob, CEM, modify=modify,
        algorithm_args={"popsize": 20, "stdev_init": 1.0, "parenthood_ratio": 0.5},
        max_inner_steps=3,
    )
    r.run(10)
    assert r.num_restarts >= 2
    assert len(seen) >= 1
    assert r.algorithm_args["popsize"] > 20


def test_external_loggers_raise_cleanly_without_package():
    from evotorch_amd.algorithms import CEM
    from evotorch_amd.logging import NeptuneLogger, SacredLogger, WandbLogger

    prob = Problem("min", sphere, solution_length=4, initial_bounds=(-1, 1), seed=1)
    s = CEM(prob, popsize=16, stdev_init=1.0, parenthood_ratio=0.5)
    for cls in (WandbLogger, NeptuneLogger):
        with pytest.raises(ImportError, match="required for this logger"):
            cls(s)
    # SacredLogger takes a sacred Run; absence of the package also raises
    with pytest.raises((ImportError, TypeError)):
        SacredLogger(s)


def test_mlflow_logger_with_stub_client(monkeypatch):
    """MlflowLogger drives any client exposing log_metric — verified with
    a stub module so the sink logic itself is exercised offline."""
    from evotorch_amd.algorithms import CEM

    calls = []

    class FakeClient:
        def log_metric(self, run_id, key, value, step=None):
            calls.append((run_id, key, value, step))

    fake = types.ModuleType("mlflow")
    fake.tracking = types.SimpleNamespace(MlflowClient=FakeClient)
    fake.active_run = lambda: types.SimpleNamespace(info=types.SimpleNamespace(run_id="r1"))
    monkeypatch.setitem(sys.modules, "mlflow", fake)

    from evotorch_amd.logging import MlflowLogger

    prob = Problem("min", sphere, solution_length=4, initial_bounds=(-1, 1), seed=2)
    s = CEM(prob, popsize=16, stdev_init=1.0, parenthood_ratio=0.5)
    MlflowLogger(s, client=FakeClient(), run=types.SimpleNamespace(info=types.SimpleNamespace(run_id="r2")))
    s.run(2)
    assert calls and all(c[0] == "r2" for c in calls)
    assert any(c[1] == "mean_eval" for c in calls)


def test_small_dsl_layers():
    """Clip/Bin/Slice/Round/Apply — the reference's utility layers, both
    directly and through the str_to_net DSL."""
    from evotorch_amd.models import str_to_net
    from evotorch_amd.models.layers import Apply, Bin, Clip, Round, Slice

    x = torch.tensor([-2.0, -0.2, 0.4, 3.0])
    torch.testing.assert_close(Clip(-1.0, 1.0)(x), torch.tensor([-1.0, -0.2, 0.4, 1.0]))
    b = Bin(-1.0, 1.0)(x)
    assert set(b.tolist()) <= {-1.0, 1.0}
    torch.testing.assert_close(Slice(1, 3)(x), torch.tensor([-0.2, 0.4]))
    torch.testing.assert_close(Round(1)(torch.tensor([0.123, 0.678])), torch.tensor([0.1, 0.7]))
    torch.testing.assert_close(Apply("tanh")(x), torch.tanh(x))
    net = str_to_net("Linear(4, 3) >> Clip(-0.5, 0.5)")
    y = net(torch.randn(4))
    assert y.abs().max() <= 0.5


def test_act_clip_layer_and_policy_export_box():
    from evotorch_amd.neuroevolution.gymne import ActClipLayer

    layer = ActClipLayer([-1.0, -2.0], [1.0, 2.0])
    out = layer(torch.tensor([[-3.0, 5.0], [0.5, -0.5]]))
    torch.testing.assert_close(out, torch.tensor([[-1.0, 2.0], [0.5, -0.5]]))


@pytest.mark.parametrize("dt", [torch.float64, torch.bfloat16, torch.float16])
def test_nondefault_problem_dtypes(dt):
    """float64 / bf16 / fp16 problems run a PGPE loop end to end (the
    reference supports arbitrary float dtypes on Problem)."""
    from evotorch_amd.algorithms import PGPE

    @vectorized
    def f(x):
        return (x.float() ** 2).sum(-1)

    p = Problem("min", f, solution_length=6, initial_bounds=(-1, 1), seed=1, dtype=dt)
    s = PGPE(p, popsize=8, center_learning_rate=0.1, stdev_learning_rate=0.1, stdev_init=0.5)
    s.run(3)
    assert s.status["iter"] == 3
    assert p.generate_batch(2).access_values().dtype == dt


def test_testing_helpers_surface():
    """The user-facing assertion helpers (reference testing.py:100-273)."""
    from evotorch_amd.testing import (
        TestingError,
        assert_allclose,
        assert_almost_between,
        assert_dtype_matches,
        assert_eachclose,
        assert_shape_matches,
    )

    a = torch.tensor([1.0, 2.0, 3.0])
    assert_allclose(a, [1.0, 2.0, 3.0], atol=1e-8)
    with pytest.raises(TestingError):
        assert_allclose(a, [1.0, 2.0, 4.0], atol=1e-3)
    assert_almost_between(a, 0.5, 3.5)
    with pytest.raises(TestingError):
        assert_almost_between(a, 1.5, 3.5)
    assert_dtype_matches(a, torch.float32)
    with pytest.raises(TestingError):
        assert_dtype_matches(a, torch.int64)
    assert_shape_matches(torch.zeros(2, 3), (2, 3))
    with pytest.raises(TestingError):
        assert_shape_matches(torch.zeros(2, 3), (3, 2))
    assert_eachclose(torch.full((4,), 7.0), 7.0)
    with pytest.raises(TestingError):
        assert_eachclose(torch.tensor([7.0, 8.0]), 7.0)
