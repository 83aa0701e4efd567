"""Plan serialization round-trip (the compiled-plan cache)."""
import numpy as np

from trtlab_amd.engine.plan_io import load_plan, save_plan
from trtlab_amd.engine.planner import DT_I8, Planner
from trtlab_amd.engine.reference import run_reference
from trtlab_amd.models import build_resnet


def test_plan_roundtrip(tmp_path):
    g = build_resnet(50, batch=1, image=64, seed=0)
    plan = Planner().compile(g)
    p = str(tmp_path / "rn50.npz")
    save_plan(plan, p)
    plan2 = load_plan(p)
    assert plan2.ops == plan.ops
    assert np.array_equal(plan2.weights, plan.weights)
    x = np.random.RandomState(2).randn(*plan.input_shape).astype(np.float32)
    assert np.array_equal(run_reference(plan, x), run_reference(plan2, x))


def test_plan_roundtrip_int8(tmp_path):
    g = build_resnet(50, batch=1, image=64, seed=0)
    plan = Planner(dtype=DT_I8).compile(g)
    p = str(tmp_path / "rn50i8.npz")
    save_plan(plan, p)
    plan2 = load_plan(p)
    assert plan2.ops == plan.ops
    x = np.random.RandomState(2).randn(*plan.input_shape).astype(np.float32)
    assert np.array_equal(run_reference(plan, x), run_reference(plan2, x))
