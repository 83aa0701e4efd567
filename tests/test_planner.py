"""Planner/IR tests (CPU): fusion correctness, liveness safety, and the
fused CPU reference vs an unfused node-level interpretation."""
import numpy as np
import pytest

from trtlab_amd.engine.calibrate import calibrate_bn
from trtlab_amd.engine.ir import Graph
from trtlab_amd.engine.planner import (EPI_SCALE_BIAS, EPI_SCALE_BIAS_ADD_RELU,
                                       EPI_SCALE_BIAS_RELU, K_CHANNEL_PAD,
                                       K_CONV, K_GEMM, Planner)
from trtlab_amd.engine.reference import run_reference
from trtlab_amd.models import build_resnet


@pytest.fixture(scope="module")
def rn50_plan():
    g = build_resnet(50, batch=2, seed=0)
    return g, Planner().compile(g)


def test_resnet50_fusion_counts(rn50_plan):
    g, plan = rn50_plan
    kinds = [d["kind"] for d in plan.ops]
    # 53 conv layers total (1 stem + 16*3 bottleneck + 4 downsample); the
    # 3 stage-1 bottleneck tails (3x3 + 1x1-expand, width 64) fuse into
    # K_BTAIL pairs, so the kernel inventory is 47 convs + 3 fused pairs
    from trtlab_amd.engine.planner import K_BTAIL
    assert kinds.count(K_BTAIL) == 3
    assert kinds.count(K_CONV) + 2 * kinds.count(K_BTAIL) == 53
    assert kinds.count(K_CHANNEL_PAD) == 1
    assert kinds.count(K_GEMM) == 1
    # every bottleneck's last conv carries the fused residual-add + relu
    # (13 standalone + 3 inside the fused tails)
    adds = [d for d in plan.ops if d.get("epi") == EPI_SCALE_BIAS_ADD_RELU]
    assert len(adds) + kinds.count(K_BTAIL) == 16
    # downsample convs are plain scale+bias
    plain = [d for d in plan.ops if d.get("epi") == EPI_SCALE_BIAS]
    assert len(plain) == 4
    relu = [d for d in plan.ops if d.get("epi") == EPI_SCALE_BIAS_RELU]
    # stem + 2 per bottleneck, minus the 3 BN+ReLU 3x3s absorbed into
    # the fused stage-1 tails
    assert len(relu) + kinds.count(K_BTAIL) == 33


def test_conv_k_padding(rn50_plan):
    g, plan = rn50_plan
    stem = next(d for d in plan.ops if d["kind"] == K_CONV)
    assert stem["C"] == 8  # padded from 3
    assert stem["KH"] == 7


def test_arena_liveness_safe(rn50_plan):
    """No two tensors with overlapping lifetimes share arena bytes."""
    g, plan = rn50_plan
    # rebuild intervals exactly as the planner does
    touched = {}

    def touch(t, i):
        s, e = touched.get(t, (i, i))
        touched[t] = (min(s, i), max(e, i))

    touch(plan.input_name, 0)
    for i, op in enumerate(plan.exec_ops):
        for t in op.inputs:
            touch(t, i)
        touch(op.output, i)
    s, e = touched[plan.output_name]
    touched[plan.output_name] = (s, len(plan.exec_ops))

    names = list(touched)
    for i, n1 in enumerate(names):
        for n2 in names[i + 1:]:
            a1, b1 = touched[n1]
            a2, b2 = touched[n2]
            if b1 < a2 or b2 < a1:
                continue
            # overlapping lifetime -> distinct offsets required
            assert plan.offsets[n1] != plan.offsets[n2], (n1, n2)


def test_fused_reference_matches_unfused_interpreter():
    """run_reference (fused exec ops, prepacked fp16 weights) must match the
    node-level fp32 interpreter within fp16-weight-quantization tolerance."""
    g = build_resnet(50, batch=2, seed=3)
    plan = Planner().compile(g)
    x = np.random.RandomState(7).randn(2, 224, 224, 3).astype(np.float32) * 0.5

    out_fused = run_reference(plan, x)

    # unfused node-level fp32 interpretation with the SAME (build-time) BN
    # stats the plan was compiled from
    import torch
    import torch.nn.functional as F

    g2 = g
    t = {g2.input_name: torch.from_numpy(x).float()}
    for n in g2.nodes:
        if n.kind == "input":
            continue
        xx = t[n.inputs[0]]
        if n.kind == "conv2d":
            w = torch.from_numpy(n.attrs["weight"])
            y = F.conv2d(xx.permute(0, 3, 1, 2), w, stride=n.attrs["stride"],
                         padding=n.attrs["padding"]).permute(0, 2, 3, 1)
        elif n.kind == "batchnorm":
            a = n.attrs
            y = (xx - torch.from_numpy(a["mean"])) / torch.sqrt(
                torch.from_numpy(a["var"]) + a["eps"])
            y = y * torch.from_numpy(a["gamma"]) + torch.from_numpy(a["beta"])
        elif n.kind == "relu":
            y = F.relu(xx)
        elif n.kind == "add":
            y = xx + t[n.inputs[1]]
        elif n.kind == "maxpool":
            y = F.max_pool2d(xx.permute(0, 3, 1, 2), n.attrs["kernel"],
                             stride=n.attrs["stride"],
                             padding=n.attrs["padding"]).permute(0, 2, 3, 1)
        elif n.kind == "gavgpool":
            nb, h, w_, c = xx.shape
            y = xx.reshape(nb, h * w_, c).mean(1)
        elif n.kind == "gemm":
            y = xx @ torch.from_numpy(n.attrs["weight"]).t()
            if n.attrs.get("bias") is not None:
                y = y + torch.from_numpy(n.attrs["bias"])
        else:
            raise AssertionError(n.kind)
        t[n.output] = y
    out_unfused = t[g2.output_name].numpy()

    # fp16 weight quantization in the fused path -> modest tolerance
    err = np.abs(out_fused - out_unfused).max()
    scale = np.abs(out_unfused).max()
    assert err / scale < 0.05, (err, scale)


def test_int8_plan_structure():
    from trtlab_amd.engine.planner import (DT_I8, K_DEQUANT, K_QUANTIZE,
                                           Planner)

    g = build_resnet(50, batch=1, image=64, seed=0)
    plan = Planner(dtype=DT_I8).compile(g)
    kinds = [d["kind"] for d in plan.ops]
    assert kinds.count(K_QUANTIZE) == 1
    assert kinds.count(K_DEQUANT) == 1
    convs = [d for d in plan.ops if d["kind"] == K_CONV]
    assert all(d["dtype"] == 2 for d in convs)
    assert all(d["C"] % 16 == 0 for d in convs)
    # residual convs carry a res_scale
    res = [d for d in convs if d["epi"] == EPI_SCALE_BIAS_ADD_RELU]
    assert len(res) == 16
    assert all(d["res_scale"] > 0 for d in res)
    # weight K padded to 128 for int8 staging
    from trtlab_amd.engine.planner import K_GEMM
    gemms = [d for d in plan.ops if d["kind"] == K_GEMM]
    assert all(d["dtype"] == 0 for d in gemms)  # head stays fp16


def test_bf16_plan_structure():
    """bf16 plans: same graph/offsets as fp16, weight blob re-encoded to
    bf16 bit patterns (decoded here via torch) while exec_ops stay fp16
    numeric for the CPU reference."""
    import torch

    from trtlab_amd.engine.planner import DT_BF16, K_GEMM, Planner

    g = build_resnet(50, batch=1, image=64, seed=0)
    plan = Planner(dtype=DT_BF16).compile(g)
    assert plan.input_dtype == "bf16"
    assert all(d["dtype"] == 1 for d in plan.ops)
    # decode the head gemm's weights from the blob and compare to exec_ops
    gd = next(d for d in plan.ops if d["kind"] == K_GEMM)
    op = next(o for o in plan.exec_ops if o.kind == K_GEMM)
    n = op.w.size
    raw = plan.weights[gd["w_off"]:gd["w_off"] + 2 * n].view(np.int16)
    dec = torch.from_numpy(raw.copy()).view(torch.bfloat16).to(
        torch.float32).numpy().reshape(op.w.shape)
    ref = op.w.astype(np.float32)
    scale = max(np.abs(ref).max(), 1e-6)
    assert np.abs(dec - ref).max() / scale < 0.01  # bf16 rounding only
    # fp32 reference executor still runs on the numeric exec_ops
    from trtlab_amd.engine.reference import run_reference

    x = np.random.RandomState(7).randn(*plan.input_shape).astype(np.float32) * 0.5
    assert np.isfinite(run_reference(plan, x)).all()


def test_fp8_plan_structure():
    from trtlab_amd.engine.planner import DT_F8, K_QUANTIZE, Planner

    g = build_resnet(50, batch=1, image=64, seed=0)
    plan = Planner(dtype=DT_F8).compile(g)
    convs = [d for d in plan.ops if d["kind"] == K_CONV]
    assert all(d["dtype"] == 3 for d in convs)
    q = next(d for d in plan.ops if d["kind"] == K_QUANTIZE)
    assert q["epi"] == 1  # fp8 format flag
    # fp8 emulation reference runs and stays sane
    from trtlab_amd.engine.reference import run_reference

    x = np.random.RandomState(5).randn(*plan.input_shape).astype(np.float32) * 0.5
    out = run_reference(plan, x)
    assert np.isfinite(out).all()


def test_resnet18_basic_block_plan():
    """Basic-block depths (18/34): residual add+relu fused into the second
    3x3 conv; reference matches the unfused interpretation elsewhere via
    the fuzz tests — here check structure + fp32 sanity."""
    from trtlab_amd.engine.planner import (EPI_SCALE_BIAS_ADD_RELU, K_CONV,
                                           Planner)
    from trtlab_amd.engine.reference import run_reference

    g = build_resnet(18, batch=1, image=64, seed=0)
    plan = Planner().compile(g)
    convs = [d for d in plan.ops if d["kind"] == K_CONV]
    fused_res = [d for d in convs if d["epi"] == EPI_SCALE_BIAS_ADD_RELU]
    assert len(fused_res) == 8  # one per basic block
    x = np.random.RandomState(2).randn(*plan.input_shape).astype(
        np.float32) * 0.5
    out = run_reference(plan, x)
    assert out.shape == (1, 1000) and np.isfinite(out).all()


def test_planner_error_paths():
    """Malformed graphs fail loudly at compile time, not on the GPU."""
    from trtlab_amd.engine.planner import Planner
    from trtlab_amd.models import build_bert

    # gemm K % 64
    g = Graph("bad")
    x = g.input((4, 100))
    with pytest.raises(AssertionError):
        g.gemm(x, np.zeros((10, 96), np.float32))  # K mismatch vs input

    g2 = Graph("bad2")
    x2 = g2.input((4, 100))
    g2.gemm(x2, np.zeros((10, 100), np.float32))
    with pytest.raises(ValueError, match="must be %64"):
        Planner().compile(g2)

    # varlen without an ids input
    g3 = build_bert(batch=1, seq=128, layers=1, seed=0)
    for n in g3.nodes:
        if n.kind == "attention":
            n.attrs["varlen"] = True
    with pytest.raises(ValueError, match="token-id"):
        Planner().compile(g3)


def test_fork_join_downsample_marking():
    """ResNet downsample convs are marked fork (side-stream) with their
    consuming residual conv marked join; pairs are strictly sequential and
    the forked inputs stay live until the join (no arena aliasing while
    the side stream reads them)."""
    from trtlab_amd.engine.planner import K_CONV, Planner
    from trtlab_amd.models import build_resnet

    plan = Planner(fork_join=True).compile(
        build_resnet(50, batch=2, image=64, seed=0))
    forks = [i for i, d in enumerate(plan.ops) if d.get("fork")]
    joins = [i for i, d in enumerate(plan.ops) if d.get("join")]
    assert len(forks) == 4 and len(joins) == 4  # one per stage transition
    seq = sorted([(i, "f") for i in forks] + [(j, "j") for j in joins])
    kinds = "".join(k for _, k in seq)
    assert kinds == "fjfjfjfj"  # strictly alternating pairs
    for f, j in zip(forks, joins):
        assert j > f + 1  # there is work to overlap
        from trtlab_amd.engine.planner import K_BTAIL
        assert plan.ops[f]["kind"] == K_CONV
        assert plan.ops[j]["kind"] in (K_CONV, K_BTAIL)
        # the join consumes the forked output as its residual
        assert plan.ops[j]["in2_off"] == plan.ops[f]["out_off"]


def test_btail_fusion_gating():
    """Bottleneck-tail fusion applies to fp16 plans only (the kernel is
    fp16), is disabled by btail_fusion=False, and never fuses when the
    3x3 output is a pinned engine output."""
    from trtlab_amd.engine.planner import DT_BF16, DT_I8, K_BTAIL, Planner
    from trtlab_amd.models import build_resnet

    g = build_resnet(50, batch=1, image=64, seed=0)
    fp16 = Planner().compile(g)
    assert sum(1 for d in fp16.ops if d["kind"] == K_BTAIL) == 3

    off = Planner(btail_fusion=False).compile(
        build_resnet(50, batch=1, image=64, seed=0))
    assert sum(1 for d in off.ops if d["kind"] == K_BTAIL) == 0

    bf16 = Planner(dtype=DT_BF16).compile(
        build_resnet(50, batch=1, image=64, seed=0))
    assert sum(1 for d in bf16.ops if d["kind"] == K_BTAIL) == 0

    i8 = Planner(dtype=DT_I8).compile(
        build_resnet(50, batch=1, image=64, seed=0))
    assert sum(1 for d in i8.ops if d["kind"] == K_BTAIL) == 0

    # pinning a 3x3 output chain as an engine output blocks that pair's
    # fusion (the intermediate must stay addressable): pin the RELU that
    # follows the first 64-wide 3x3 conv (the tensor the fusion would
    # otherwise swallow into LDS)
    g2 = build_resnet(50, batch=1, image=64, seed=0)
    first3 = next(n for n in g2.nodes
                  if n.kind == "conv2d" and
                  n.attrs["weight"].shape == (64, 64, 3, 3))
    relu = next(n for n in g2.nodes
                if n.kind == "relu" and len(n.inputs) == 1 and
                any(m.kind == "batchnorm" and m.inputs[0] == first3.output
                    and n.inputs[0] == m.output for m in g2.nodes))
    g2.mark_output(relu.output)
    pinned = Planner().compile(g2)
    assert sum(1 for d in pinned.ops if d["kind"] == K_BTAIL) == 2
