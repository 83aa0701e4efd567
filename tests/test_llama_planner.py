"""LLaMA-family (RMSNorm + RoPE + SwiGLU) CPU tests: plan structure and
reference-executor numerics vs an independent torch implementation."""
import numpy as np
import pytest
import torch

from trtlab_amd.engine.planner import (K_GEMM, K_RMSNORM, K_ROPE,
                                       K_SILU_MUL, Planner)
from trtlab_amd.engine.reference import run_reference
from trtlab_amd.models import build_llama


def _torch_llama(g, ids, seq, heads, theta=10000.0):
    """Independent oracle straight from the IR node attrs."""
    nodes = {n.name: n for n in g.nodes}
    emb = nodes["embed"]
    tok = torch.from_numpy(emb.attrs["tok"])
    h = tok[torch.from_numpy(ids).long()]
    hidden = h.shape[-1]
    hd = hidden // heads
    half = hd // 2
    m = h.shape[0]
    B = m // seq

    def rms(x, gname):
        gam = torch.from_numpy(nodes[gname].attrs["gamma"])
        return x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-5) * gam

    li = 0
    while f"l{li}_qkv" in nodes:
        x = rms(h, f"l{li}_rms1")
        qkv = x @ torch.from_numpy(nodes[f"l{li}_qkv"].attrs["weight"]).t()
        # rope on q/k
        pos = (torch.arange(m) % seq).double()
        d = torch.arange(half, dtype=torch.float64)
        ang = pos[:, None] * theta ** (-2.0 * d / hd)
        cos, sin = torch.cos(ang).float(), torch.sin(ang).float()
        qkv = qkv.clone()
        for blk in range(2):
            v = qkv[:, blk * hidden:(blk + 1) * hidden].reshape(m, heads, hd)
            x0, x1 = v[..., :half].clone(), v[..., half:].clone()
            v[..., :half] = x0 * cos[:, None] - x1 * sin[:, None]
            v[..., half:] = x0 * sin[:, None] + x1 * cos[:, None]
        q, k, v = (qkv[:, i * hidden:(i + 1) * hidden]
                   .reshape(B, seq, heads, hd).permute(0, 2, 1, 3)
                   for i in range(3))
        sc = (q @ k.transpose(-1, -2)) / hd ** 0.5
        cm = torch.arange(seq)[None, :] > torch.arange(seq)[:, None]
        sc = sc.masked_fill(cm[None, None], float("-inf"))
        att = (torch.softmax(sc, -1) @ v).permute(0, 2, 1, 3).reshape(
            m, hidden)
        proj = att @ torch.from_numpy(
            nodes[f"l{li}_proj"].attrs["weight"]).t()
        h = h + proj
        x = rms(h, f"l{li}_rms2")
        gate = x @ torch.from_numpy(
            nodes[f"l{li}_gate"].attrs["weight"]).t()
        up = x @ torch.from_numpy(nodes[f"l{li}_up"].attrs["weight"]).t()
        ff = torch.nn.functional.silu(gate) * up
        down = ff @ torch.from_numpy(
            nodes[f"l{li}_down"].attrs["weight"]).t()
        h = h + down
        li += 1
    return rms(h, "rms_f").numpy()


def test_llama_plan_structure():
    g = build_llama(batch=1, seq=128, hidden=1024, layers=2, heads=8,
                    seed=0)
    plan = Planner().compile(g)
    kinds = [d["kind"] for d in plan.ops]
    assert kinds.count(K_RMSNORM) == 2 * 2 + 1
    assert kinds.count(K_ROPE) == 2
    assert kinds.count(K_SILU_MUL) == 2
    assert kinds.count(K_GEMM) == 2 * 5  # qkv, proj, gate, up, down
    # rope output aliases its input in the arena (in-place rotation)
    for op, d in zip(plan.exec_ops, plan.ops):
        if d["kind"] == K_ROPE:
            assert plan.offsets[op.output] == plan.offsets[op.inputs[0]]
            assert d["HD"] == 128


def test_llama_reference_matches_torch():
    g = build_llama(batch=2, seq=64, hidden=512, layers=2, heads=4, seed=1)
    plan = Planner().compile(g)
    ids = np.random.RandomState(2).randint(
        1, 30000, size=plan.input_shape).astype(np.int32)
    out = run_reference(plan, ids)
    ref = _torch_llama(g, ids, seq=64, heads=4)
    err = np.abs(out - ref).max() / max(np.abs(ref).max(), 1e-6)
    assert err < 2e-3, err
