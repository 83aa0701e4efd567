"""BERT IR/planner CPU tests."""
import numpy as np

from trtlab_amd.engine.planner import (EPI_BIAS, EPI_BIAS_GELU, K_ATTENTION,
                                       K_ADD_LAYERNORM, K_GEMM, Planner)
from trtlab_amd.models import build_bert


def test_bert_plan_structure():
    g = build_bert(batch=2, seq=128, layers=2, seed=0)
    plan = Planner().compile(g)
    kinds = [d["kind"] for d in plan.ops]
    assert kinds.count(K_GEMM) == 2 * 4      # qkv, proj, ff1, ff2 per layer
    assert kinds.count(K_ATTENTION) == 2
    assert kinds.count(K_ADD_LAYERNORM) == 2 * 2
    gelus = [d for d in plan.ops if d["kind"] == K_GEMM and d["epi"] == EPI_BIAS_GELU]
    assert len(gelus) == 2                   # ff1 fused with gelu
    att = next(d for d in plan.ops if d["kind"] == K_ATTENTION)
    assert att["B"] == 2 and att["S"] == 128 and att["NH"] == 12 and att["HD"] == 64


def test_bert_reference_runs():
    from trtlab_amd.engine.reference import run_reference

    g = build_bert(batch=1, seq=128, layers=1, seed=1)
    plan = Planner().compile(g)
    x = np.random.RandomState(3).randn(*plan.input_shape).astype(np.float32)
    out = run_reference(plan, x)
    assert out.shape == plan.output_shape
    assert np.isfinite(out).all()
    assert np.abs(out).max() < 50  # layernorm keeps things bounded
