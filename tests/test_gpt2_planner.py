"""GPT-2-family (pre-LN, causal attention) plan + causality tests."""
import numpy as np
import pytest

from trtlab_amd.engine.planner import K_ATTENTION, Planner
from trtlab_amd.engine.reference import run_reference
from trtlab_amd.models import build_gpt2


def test_gpt2_plan_compiles_and_runs():
    g = build_gpt2(batch=2, seq=128, layers=2, seed=0)
    plan = Planner().compile(g)
    atts = [d for d in plan.ops if d["kind"] == K_ATTENTION]
    assert len(atts) == 2 and all(d["causal"] == 1 for d in atts)
    x = np.random.RandomState(1).randn(*plan.input_shape).astype(
        np.float32) * 0.5
    out = run_reference(plan, x)
    assert out.shape == plan.output_shape and np.isfinite(out).all()


def test_gpt2_causality_invariant():
    """Output at position i must not depend on positions > i."""
    g = build_gpt2(batch=1, seq=128, layers=2, seed=0)
    plan = Planner().compile(g)
    rng = np.random.RandomState(2)
    x = rng.randn(*plan.input_shape).astype(np.float32) * 0.5
    y1 = run_reference(plan, x)
    x2 = x.copy()
    x2[64:] = rng.randn(64, x.shape[1]) * 0.5  # perturb the future
    y2 = run_reference(plan, x2)
    assert np.allclose(y1[:64], y2[:64], atol=1e-5)
    assert not np.allclose(y1[64:], y2[64:], atol=1e-3)
