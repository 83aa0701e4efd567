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


def test_plan_roundtrip_mx_and_varlen(tmp_path):
    """New op kinds (MX gemms, seqlens) survive the plan cache byte-for-
    byte: op dicts, second weight slab offsets, input dtype."""
    import numpy as np

    from trtlab_amd.engine.plan_io import load_plan, save_plan
    from trtlab_amd.engine.planner import DT_MX8, Planner
    from trtlab_amd.engine.reference import run_reference
    from trtlab_amd.models import build_bert

    plan = Planner(dtype=DT_MX8).compile(
        build_bert(batch=2, seq=128, layers=1, seed=0))
    p = tmp_path / "mx.plan"
    save_plan(plan, str(p))
    plan2 = load_plan(str(p))
    assert plan2.ops == plan.ops
    assert np.array_equal(plan2.weights, plan.weights)
    assert plan2.dtype == plan.dtype

    vplan = Planner().compile(build_bert(batch=2, seq=128, layers=1, seed=0,
                                         embeddings=True, varlen=True))
    p2 = tmp_path / "vl.plan"
    save_plan(vplan, str(p2))
    vplan2 = load_plan(str(p2))
    assert vplan2.ops == vplan.ops and vplan2.input_dtype == "i32"
