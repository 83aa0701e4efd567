"""HF-safetensors checkpoint interop: the loader's name mapping + rope/
rmsnorm/SwiGLU conventions are validated against an INDEPENDENT torch
implementation of HF LLaMA semantics (rotate_half rotary, rms_norm_eps,
nn.Linear x @ W^T), on a synthetic checkpoint written with safetensors."""
import json
import os

import numpy as np
import pytest
import torch

from trtlab_amd.engine.planner import Planner
from trtlab_amd.engine.reference import run_reference
from trtlab_amd.models import build_llama_from_safetensors

H, LAYERS, HEADS, VOCAB, INTER = 256, 2, 2, 300, 448
EPS, THETA = 1e-6, 10000.0


def _mk_checkpoint(tmpdir, kvh=HEADS):
    from safetensors.numpy import save_file

    rng = np.random.RandomState(7)
    hd = H // HEADS
    kv_rows = kvh * hd

    def w(o, i):
        return (rng.randn(o, i) / np.sqrt(i)).astype(np.float32)

    st = {"model.embed_tokens.weight":
          (rng.randn(VOCAB, H) * 0.05).astype(np.float32),
          "model.norm.weight":
          rng.uniform(0.9, 1.1, H).astype(np.float32)}
    for li in range(LAYERS):
        p = f"model.layers.{li}."
        st[p + "input_layernorm.weight"] = \
            rng.uniform(0.9, 1.1, H).astype(np.float32)
        st[p + "post_attention_layernorm.weight"] = \
            rng.uniform(0.9, 1.1, H).astype(np.float32)
        for nm, shape in (("self_attn.q_proj", (H, H)),
                          ("self_attn.k_proj", (kv_rows, H)),
                          ("self_attn.v_proj", (kv_rows, H)),
                          ("self_attn.o_proj", (H, H)),
                          ("mlp.gate_proj", (INTER, H)),
                          ("mlp.up_proj", (INTER, H)),
                          ("mlp.down_proj", (H, INTER))):
            st[p + nm + ".weight"] = w(*shape)
    save_file(st, os.path.join(tmpdir, "model.safetensors"))
    with open(os.path.join(tmpdir, "config.json"), "w") as f:
        json.dump({"num_attention_heads": HEADS,
                   "num_key_value_heads": kvh,
                   "rope_theta": THETA, "rms_norm_eps": EPS}, f)
    return st


def _hf_forward(st, ids, seq, kvh=HEADS):
    """Independent HF-semantics oracle (fp32 torch; GQA when
    kvh < HEADS — query head h attends kv head h // (HEADS//kvh))."""
    hd = H // HEADS
    grp = HEADS // kvh

    def rms(x, g):
        v = x / torch.sqrt((x * x).mean(-1, keepdim=True) + EPS)
        return v * torch.from_numpy(g)

    def lin(x, wname):
        return x @ torch.from_numpy(st[wname]).t()

    def rope(x, pos):  # x [S, heads, hd] — HF rotate_half
        half = hd // 2
        inv = THETA ** (-torch.arange(half, dtype=torch.float64) * 2 / hd)
        ang = pos[:, None].double() * inv[None, :]
        cos = torch.cos(ang).float()[:, None, :]
        sin = torch.sin(ang).float()[:, None, :]
        x1, x2 = x[..., :half], x[..., half:]
        return torch.cat([x1 * cos - x2 * sin, x1 * sin + x2 * cos], -1)

    h = torch.from_numpy(st["model.embed_tokens.weight"])[ids]  # [S, H]
    pos = torch.arange(seq)
    for li in range(LAYERS):
        p = f"model.layers.{li}."
        x = rms(h, st[p + "input_layernorm.weight"])
        q = lin(x, p + "self_attn.q_proj.weight").view(seq, HEADS, hd)
        k = lin(x, p + "self_attn.k_proj.weight").view(seq, kvh, hd)
        v = lin(x, p + "self_attn.v_proj.weight").view(seq, kvh, hd)
        q, k = rope(q, pos), rope(k, pos)
        att = torch.zeros(seq, HEADS, hd)
        mask = torch.tril(torch.ones(seq, seq, dtype=torch.bool))
        for hh in range(HEADS):
            kk = k[:, hh // grp]
            vv = v[:, hh // grp]
            sc = (q[:, hh] @ kk.t()) / np.sqrt(hd)
            sc = sc.masked_fill(~mask, float("-inf"))
            att[:, hh] = torch.softmax(sc, -1) @ vv
        h = h + lin(att.reshape(seq, H), p + "self_attn.o_proj.weight")
        x = rms(h, st[p + "post_attention_layernorm.weight"])
        gate = lin(x, p + "mlp.gate_proj.weight")
        up = lin(x, p + "mlp.up_proj.weight")
        ff = torch.nn.functional.silu(gate) * up
        h = h + lin(ff, p + "mlp.down_proj.weight")
    return rms(h, st["model.norm.weight"]).numpy()


def test_safetensors_llama_matches_hf_semantics(tmp_path):
    st = _mk_checkpoint(str(tmp_path))
    seq = 24
    g = build_llama_from_safetensors(str(tmp_path), batch=1, seq=seq)
    plan = Planner().compile(g)
    ids = np.random.RandomState(3).randint(0, VOCAB, seq).astype(np.int32)
    out = run_reference(plan, ids)
    ref = _hf_forward(st, torch.from_numpy(ids).long(), seq)
    err = np.abs(out - ref).max() / max(np.abs(ref).max(), 1e-6)
    assert err < 2e-3, err


def test_safetensors_llama_gqa_matches_hf_semantics(tmp_path):
    """GQA (1 kv head serving 2 query heads) loads via exact kv-head
    replication and matches the true grouped-attention oracle."""
    st = _mk_checkpoint(str(tmp_path), kvh=1)
    seq = 20
    g = build_llama_from_safetensors(str(tmp_path), batch=1, seq=seq)
    plan = Planner().compile(g)
    ids = np.random.RandomState(5).randint(0, VOCAB, seq).astype(np.int32)
    out = run_reference(plan, ids)
    ref = _hf_forward(st, torch.from_numpy(ids).long(), seq, kvh=1)
    err = np.abs(out - ref).max() / max(np.abs(ref).max(), 1e-6)
    assert err < 2e-3, err


def test_safetensors_llama_decode_session_compat(tmp_path):
    """The loaded graph exposes the node names DecodeSession's llama
    recipe extracts (l{i}_rms1/qkv/rope/att/proj/rms2/gate/up/down,
    rms_f) — checked structurally on CPU; the GPU decode test covers the
    numerics for the same recipe."""
    _mk_checkpoint(str(tmp_path))
    g = build_llama_from_safetensors(str(tmp_path), batch=2, seq=32)
    names = {n.name for n in g.nodes}
    for li in range(LAYERS):
        for suff in ("rms1", "qkv", "rope", "att", "proj", "rms2",
                     "gate", "up", "down"):
            assert f"l{li}_{suff}" in names, suff
    assert "rms_f" in names and "embed" in names
