"""Multi-binding engine I/O CPU tests (reference Bindings carve N host +
device addresses per model, trtlab/tensorrt/bindings.h:60-120): N named
inputs/outputs flow IR -> planner -> plan -> CPU reference. The GPU-side
counterpart (captured 3-input/2-output forward) is tests/test_engine_gpu.py.
"""
import numpy as np
import pytest

from trtlab_amd.engine.ir import Graph
from trtlab_amd.engine.planner import Planner
from trtlab_amd.engine.reference import run_reference
from trtlab_amd.models import build_bert


def _three_in_two_out():
    """x @ W1 (+y residual add) -> h (marked output); h @ W2 + z -> out."""
    rng = np.random.RandomState(0)
    g = Graph("multi_io")
    x = g.input((32, 64), name="x")
    y = g.input((32, 128), name="y")
    z = g.input((32, 64), name="z")
    w1 = (rng.randn(128, 64) * 0.1).astype(np.float32)
    h0 = g.gemm(x, w1, (rng.randn(128) * 0.1).astype(np.float32), name="g1")
    h = g.add(h0, y, name="mid")
    g.mark_output(h)
    w2 = (rng.randn(64, 128) * 0.1).astype(np.float32)
    o0 = g.gemm(h, w2, (rng.randn(64) * 0.1).astype(np.float32), name="g2")
    g.add(o0, z, name="out")
    return g


def test_plan_carries_all_bindings():
    plan = Planner().compile(_three_in_two_out())
    assert [b["name"] for b in plan.inputs] == ["x", "y", "z"]
    names = [b["name"] for b in plan.outputs]
    assert "mid" in names and "out" in names
    # every binding maps to a distinct arena region of the right size
    for b in plan.inputs + plan.outputs:
        assert b["off"] >= 0 and b["bytes"] > 0
        assert b["off"] + b["bytes"] <= plan.arena_bytes
    # legacy single-binding fields mirror binding [0]
    assert plan.inputs[0]["off"] == plan.input_off
    assert plan.outputs[0]["bytes"] == plan.output_bytes


def test_marked_output_not_clobbered_by_arena_reuse():
    """mid is consumed early but marked as an output: its storage must not
    be reused by later tensors (liveness extends to the end)."""
    plan = Planner().compile(_three_in_two_out())
    mid = next(b for b in plan.outputs if b["name"] == "mid")
    out = next(b for b in plan.outputs if b["name"] == "out")
    lo1, hi1 = mid["off"], mid["off"] + mid["bytes"]
    lo2, hi2 = out["off"], out["off"] + out["bytes"]
    assert hi1 <= lo2 or hi2 <= lo1, "output bindings overlap in the arena"


def test_reference_multi_input():
    plan = Planner().compile(_three_in_two_out())
    rng = np.random.RandomState(1)
    feeds = {"x": rng.randn(32, 64).astype(np.float32),
             "y": rng.randn(32, 128).astype(np.float32),
             "z": rng.randn(32, 64).astype(np.float32)}
    all_t = run_reference(plan, feeds, return_all=True)
    w1 = plan.exec_ops[0].w.astype(np.float32)
    b1 = plan.exec_ops[0].bias
    mid = feeds["x"] @ w1.T + b1 + feeds["y"]
    assert np.allclose(all_t["mid"], mid, atol=1e-3)
    assert np.isfinite(all_t["out"]).all()


def test_bert_three_binding_plan():
    """BERT with REAL (ids, mask, segments) bindings: 3 i32 inputs; the
    mask drives the varlen key masking; segments add the token-type table."""
    g = build_bert(batch=2, seq=128, layers=1, seed=0, embeddings=True,
                   varlen=True, segments=True, mask_input=True)
    plan = Planner().compile(g)
    assert [b["name"] for b in plan.inputs] == [
        "token_ids", "segment_ids", "attention_mask"]
    assert all(b["dtype"] == "i32" for b in plan.inputs)
    # seqlens reads the attention_mask binding, not the token ids
    from trtlab_amd.engine.planner import K_SEQLENS

    sl = next(o for o in plan.exec_ops if o.kind == K_SEQLENS)
    assert sl.inputs == ["attention_mask"]
    # embedding consumes segment_ids as its second input
    from trtlab_amd.engine.planner import K_EMBEDDING

    emb = next(o for o in plan.exec_ops if o.kind == K_EMBEDDING)
    assert emb.inputs == ["token_ids", "segment_ids"]


def test_bert_three_binding_reference_masking():
    """Padded positions (mask=0) must not influence valid rows, and the
    segment table must shift embeddings: compare against a mask-free run."""
    seq, b = 128, 2
    g = build_bert(batch=b, seq=seq, layers=1, seed=0, embeddings=True,
                   varlen=True, segments=True, mask_input=True)
    plan = Planner().compile(g)
    rng = np.random.RandomState(5)
    ids = rng.randint(1, 30000, size=(b * seq,)).astype(np.int32)
    segs = np.zeros(b * seq, np.int32)
    segs[seq // 2:seq] = 1  # second half of sequence 0 is segment B
    mask = np.ones(b * seq, np.int32)
    full = run_reference(plan, {"token_ids": ids, "segment_ids": segs,
                                "attention_mask": mask})
    # now mask off the tail of sequence 1
    mask2 = mask.copy()
    valid = 77
    mask2[seq + valid:] = 0
    part = run_reference(plan, {"token_ids": ids, "segment_ids": segs,
                                "attention_mask": mask2})
    fullv = full.reshape(b, seq, -1)
    partv = part.reshape(b, seq, -1)
    # sequence 0 untouched by sequence 1's padding
    assert np.allclose(partv[0], fullv[0], atol=1e-4)
    # sequence 1's valid rows differ (fewer keys attended)
    assert not np.allclose(partv[1, :valid], fullv[1, :valid], atol=1e-3)


def test_plan_io_roundtrip_bindings(tmp_path):
    from trtlab_amd.engine.plan_io import load_plan, save_plan

    plan = Planner().compile(_three_in_two_out())
    p = tmp_path / "m.plan"
    save_plan(plan, str(p))
    plan2 = load_plan(str(p))
    assert plan2.inputs == plan.inputs
    assert plan2.outputs == plan.outputs
