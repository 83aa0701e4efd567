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


def test_varlen_plan_structure_and_noop_equivalence():
    """varlen BERT: a seqlens op is scheduled once, every attention op reads
    it; with full-length (pad-free) ids the masked plan must match the
    unmasked plan exactly."""
    import numpy as np

    from trtlab_amd.engine.planner import K_ATTENTION, K_SEQLENS, Planner
    from trtlab_amd.engine.reference import run_reference
    from trtlab_amd.models import build_bert

    g = build_bert(batch=2, seq=128, layers=2, seed=0, embeddings=True,
                   varlen=True)
    plan = Planner().compile(g)
    kinds = [d["kind"] for d in plan.ops]
    assert kinds.count(K_SEQLENS) == 1
    atts = [d for d in plan.ops if d["kind"] == K_ATTENTION]
    assert len(atts) == 2 and all("in2_off" in d for d in atts)

    rng = np.random.RandomState(3)
    ids_full = rng.randint(1, 30522, 256).astype(np.int32)  # no pad tokens
    out_v = run_reference(plan, ids_full)

    g2 = build_bert(batch=2, seq=128, layers=2, seed=0, embeddings=True)
    plan2 = Planner().compile(g2)
    out_p = run_reference(plan2, ids_full)
    assert np.allclose(out_v, out_p), "full-length varlen must be a no-op"

    # padded batch: lens derived correctly, masking changes the result
    ids_pad = ids_full.copy().reshape(2, 128)
    ids_pad[1, 57:] = 0  # right-pad sequence 1 to length 57
    ids_pad = ids_pad.reshape(-1)
    all_t = run_reference(plan, ids_pad, return_all=True)
    assert list(all_t["_seqlens"].astype(int)) == [128, 57]
    out_masked = all_t[plan.output_name]
    out_unmasked = run_reference(plan2, ids_pad)
    assert np.isfinite(out_masked).all()
    # sequence 0 rows see a fully-valid batch -> unaffected by masking of
    # sequence 1 (attention never crosses sequences)
    assert np.allclose(out_masked[:128], out_unmasked[:128], atol=1e-5)
    # sequence 1 valid rows must change (padded keys no longer attended)
    assert not np.allclose(out_masked[128:128 + 57],
                           out_unmasked[128:128 + 57], atol=1e-3)


def test_mxfp4_plan_structure():
    """DT_MX4 lowers every BERT GEMM to quantize + MXFP4 scaled-MFMA GEMM;
    the fp32 reference emulation stays close to the fp16 plan."""
    import numpy as np

    from trtlab_amd.engine.planner import (DT_MX4, K_GEMM, K_GEMM_MX4,
                                           K_QUANT_MX4, Planner)
    from trtlab_amd.engine.reference import run_reference
    from trtlab_amd.models import build_bert

    g = build_bert(batch=2, seq=128, layers=2, seed=0)
    plan = Planner(dtype=DT_MX4).compile(g)
    kinds = [d["kind"] for d in plan.ops]
    assert kinds.count(K_GEMM_MX4) == 8  # 4 gemms x 2 layers
    # producer fusion: quants whose row comes from a layernorm are folded
    # INTO that LN (epi = 4 marks the MX-emitting LN); only rows produced
    # by non-LN ops (graph input, attention, ff1-gelu) keep a standalone
    # quantize — 2-layer BERT: 3 fused (l0.ff1, l1.qkv, l1.ff1), 5 left
    assert kinds.count(K_QUANT_MX4) == 5
    fused_ln = [d for d in plan.ops
                if d["kind"] in (5, 6) and d.get("epi") == 4]
    assert len(fused_ln) == 3
    assert all(d["out2_off"] >= 0 and d["out3_off"] >= 0 for d in fused_ln)
    assert kinds.count(K_GEMM) == 0
    mx = [d for d in plan.ops if d["kind"] == K_GEMM_MX4]
    assert all(d["w2_off"] >= 0 and "in2_off" in d for d in mx)

    x = np.random.RandomState(3).randn(*plan.input_shape).astype(
        np.float32) * 0.5
    out = run_reference(plan, x)
    assert np.isfinite(out).all()
    g2 = build_bert(batch=2, seq=128, layers=2, seed=0)
    ref16 = run_reference(Planner().compile(g2), x)
    corr = np.corrcoef(out.ravel(), ref16.ravel())[0, 1]
    assert corr > 0.95, corr  # fp4 weights+activations: coarse but sane


def test_mxfp8_plan_structure():
    """DT_MX8: same lowering with e4m3 elements (1 byte/elem codes)."""
    import numpy as np

    from trtlab_amd.engine.planner import (DT_MX8, K_GEMM, K_GEMM_MX8,
                                           K_QUANT_MX8, Planner)
    from trtlab_amd.engine.reference import run_reference
    from trtlab_amd.models import build_bert

    g = build_bert(batch=2, seq=128, layers=1, seed=0)
    plan = Planner(dtype=DT_MX8).compile(g)
    kinds = [d["kind"] for d in plan.ops]
    assert kinds.count(K_GEMM_MX8) == 4
    assert kinds.count(K_QUANT_MX8) == 3  # ff1's quant fused into add_ln
    assert sum(1 for d in plan.ops
               if d["kind"] in (5, 6) and d.get("epi") == 8) == 1
    assert kinds.count(K_GEMM) == 0
    x = np.random.RandomState(3).randn(*plan.input_shape).astype(
        np.float32) * 0.5
    out = run_reference(plan, x)
    g2 = build_bert(batch=2, seq=128, layers=1, seed=0)
    ref16 = run_reference(Planner().compile(g2), x)
    corr = np.corrcoef(out.ravel(), ref16.ravel())[0, 1]
    assert corr > 0.99, corr  # fp8 elements: much tighter than fp4
