"""Vision Transformer family: conv patch embedding + view/constant ops +
encoder attention + mean-pool head, validated against an independent
plain-torch ViT on CPU and the native engine on GPU."""
import numpy as np
import pytest
import torch
import torch.nn.functional as F

from trtlab_amd.engine.planner import K_CONST, K_VIEW, Planner
from trtlab_amd.engine.reference import run_reference
from trtlab_amd.models import build_vit


def _torch_vit(g, x):
    """Independent oracle using the weights stored in the IR graph."""
    nodes = {n.name: n for n in g.nodes}
    xt = torch.from_numpy(x)
    pw = torch.from_numpy(nodes["patch_embed"].attrs["weight"])
    stride = nodes["patch_embed"].attrs["stride"]
    h = F.conv2d(xt.permute(0, 3, 1, 2), pw,
                 stride=stride).permute(0, 2, 3, 1)
    B, gr, _, H = h.shape
    S = gr * gr
    h = h.reshape(B * S, H)
    h = h + torch.from_numpy(nodes["pos_embed"].attrs["value"])

    def lnorm(v, n):
        a = nodes[n].attrs
        mu = v.mean(-1, keepdim=True)
        var = v.var(-1, unbiased=False, keepdim=True)
        return ((v - mu) / torch.sqrt(var + 1e-5) *
                torch.from_numpy(a["gamma"]) + torch.from_numpy(a["beta"]))

    def gm(v, n):
        a = nodes[n].attrs
        r = v @ torch.from_numpy(a["weight"]).t()
        return r + torch.from_numpy(a["bias"]) if a["bias"] is not None \
            else r

    li = 0
    while f"l{li}_ln1" in nodes:
        NH = nodes[f"l{li}_att"].attrs["heads"]
        hd = H // NH
        xn = lnorm(h, f"l{li}_ln1")
        qkv = gm(xn, f"l{li}_qkv").reshape(B, S, 3, NH, hd)
        att = torch.zeros(B, S, NH, hd)
        for b in range(B):
            for hh in range(NH):
                sc = (qkv[b, :, 0, hh] @ qkv[b, :, 1, hh].t()) / np.sqrt(hd)
                att[b, :, hh] = torch.softmax(sc, -1) @ qkv[b, :, 2, hh]
        h = h + gm(att.reshape(B * S, H), f"l{li}_proj")
        xn = lnorm(h, f"l{li}_ln2")
        ff = gm(xn, f"l{li}_ff1")
        ff = 0.5 * ff * (1 + torch.tanh(
            np.sqrt(2 / np.pi) * (ff + 0.044715 * ff ** 3)))
        h = h + gm(ff, f"l{li}_ff2")
        li += 1
    h = lnorm(h, "ln_f")
    return gm(h.reshape(B, S, H).mean(1), "head").numpy()


def test_vit_reference_matches_torch_oracle():
    g = build_vit(batch=2, image=64, patch=16, hidden=256, layers=2,
                  heads=4, classes=10, seed=0)
    plan = Planner().compile(g)
    kinds = [d["kind"] for d in plan.ops]
    assert kinds.count(K_VIEW) == 2  # token flatten + pool reshape
    assert kinds.count(K_CONST) == 1  # position embedding
    x = (np.random.RandomState(0).randn(*plan.input_shape) * 0.5).astype(
        np.float32)
    out = run_reference(plan, x)
    ref = _torch_vit(g, x)
    err = np.abs(out - ref).max() / max(np.abs(ref).max(), 1e-6)
    assert err < 2e-3, err


def test_vit_view_is_zero_copy():
    """Views alias their source in the arena (no extra bytes, no op)."""
    g = build_vit(batch=1, image=32, patch=16, hidden=256, layers=1,
                  heads=4, classes=10, seed=1)
    plan = Planner().compile(g)
    views = [(op, d) for op, d in zip(plan.exec_ops, plan.ops)
             if d["kind"] == K_VIEW]
    assert views
    for op, d in views:
        assert d["in_off"] == d["out_off"]  # aliased, not copied


def test_vit_plan_roundtrip(tmp_path):
    import os

    from trtlab_amd.engine.plan_io import load_plan, save_plan

    g = build_vit(batch=1, image=32, patch=16, hidden=256, layers=1,
                  heads=4, classes=10, seed=2)
    plan = Planner().compile(g)
    pth = os.path.join(str(tmp_path), "vit.npz")
    save_plan(plan, pth)
    p2 = load_plan(pth)
    x = (np.random.RandomState(1).randn(*plan.input_shape) * 0.5).astype(
        np.float32)
    a = run_reference(plan, x)
    b = run_reference(p2, x)
    assert np.abs(a - b).max() < 1e-6


@pytest.mark.gpu
def test_vit_engine_matches_reference():
    """ViT end-to-end on the captured engine (conv patch embed + view
    aliases + device constant + encoder attention + gavgpool head)."""
    from trtlab_amd.engine.runtime import NativeEngine

    g = build_vit(batch=2, image=64, patch=16, hidden=256, layers=2,
                  heads=4, classes=10, seed=0)
    plan = Planner().compile(g)
    ctx = NativeEngine(plan).create_context(capture=True)
    x = (np.random.RandomState(3).randn(*plan.input_shape) * 0.5).astype(
        np.float32)
    out = ctx.infer(x).astype(np.float32)
    ref = run_reference(plan, x)
    err = np.abs(out - ref).max() / max(np.abs(ref).max(), 1e-6)
    assert err < 0.08, err


def test_llama_plan_roundtrip(tmp_path):
    """LLaMA plans (rope arena aliasing, rmsnorm/silu ops) survive
    save_plan/load_plan byte-exactly."""
    import os

    from trtlab_amd.engine.plan_io import load_plan, save_plan
    from trtlab_amd.models import build_llama

    g = build_llama(batch=1, seq=32, hidden=256, layers=1, heads=2,
                    seed=0, vocab=400)
    plan = Planner().compile(g)
    pth = os.path.join(str(tmp_path), "llama.npz")
    save_plan(plan, pth)
    p2 = load_plan(pth)
    ids = np.random.RandomState(2).randint(
        0, 400, plan.input_shape).astype(np.int32)
    a = run_reference(plan, ids)
    b = run_reference(p2, ids)
    assert np.abs(a - b).max() < 1e-6
