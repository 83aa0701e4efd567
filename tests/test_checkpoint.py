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


def test_safetensors_gpt2_matches_hf_semantics(tmp_path):
    """GPT-2 loader: Conv1D transposition + tanh-GELU + learned
    positions vs an independent HF-semantics oracle."""
    from safetensors.numpy import save_file

    from trtlab_amd.models import build_gpt2_from_safetensors

    Hh, LAY, HD, V, P = 128, 2, 2, 200, 64  # head_dim 64
    rng = np.random.RandomState(11)
    st = {"transformer.wte.weight":
          (rng.randn(V, Hh) * 0.05).astype(np.float32),
          "transformer.wpe.weight":
          (rng.randn(P, Hh) * 0.05).astype(np.float32),
          "transformer.ln_f.weight":
          rng.uniform(0.9, 1.1, Hh).astype(np.float32),
          "transformer.ln_f.bias":
          (rng.randn(Hh) * 0.02).astype(np.float32)}
    for li in range(LAY):
        p = f"transformer.h.{li}."
        for nm, shp in (("ln_1.weight", (Hh,)), ("ln_1.bias", (Hh,)),
                        ("ln_2.weight", (Hh,)), ("ln_2.bias", (Hh,)),
                        ("attn.c_attn.weight", (Hh, 3 * Hh)),
                        ("attn.c_attn.bias", (3 * Hh,)),
                        ("attn.c_proj.weight", (Hh, Hh)),
                        ("attn.c_proj.bias", (Hh,)),
                        ("mlp.c_fc.weight", (Hh, 4 * Hh)),
                        ("mlp.c_fc.bias", (4 * Hh,)),
                        ("mlp.c_proj.weight", (4 * Hh, Hh)),
                        ("mlp.c_proj.bias", (Hh,))):
            scale = 0.02 if nm.endswith("bias") else 1 / np.sqrt(shp[0])
            st[p + nm] = (rng.randn(*shp) * scale).astype(np.float32)
        st[p + "ln_1.weight"] = rng.uniform(0.9, 1.1, Hh).astype(np.float32)
        st[p + "ln_2.weight"] = rng.uniform(0.9, 1.1, Hh).astype(np.float32)
    save_file(st, os.path.join(str(tmp_path), "model.safetensors"))
    with open(tmp_path / "config.json", "w") as f:
        json.dump({"n_head": HD}, f)

    seq = 16
    g = build_gpt2_from_safetensors(str(tmp_path), batch=1, seq=seq)
    plan = Planner().compile(g)
    ids = np.random.RandomState(4).randint(0, V, seq).astype(np.int32)
    out = run_reference(plan, ids)

    # --- independent HF-GPT2 oracle (Conv1D: x @ W) ---
    def t(k):
        return torch.from_numpy(st["transformer." + k])

    def ln(x, w, b):
        mu = x.mean(-1, keepdim=True)
        var = x.var(-1, unbiased=False, keepdim=True)
        return (x - mu) / torch.sqrt(var + 1e-5) * t(w) + t(b)

    def gelu_new(x):
        return 0.5 * x * (1 + torch.tanh(
            np.sqrt(2 / np.pi) * (x + 0.044715 * x ** 3)))

    hid = t("wte.weight")[torch.from_numpy(ids).long()] + \
        t("wpe.weight")[:seq]
    hd = Hh // HD
    mask = torch.tril(torch.ones(seq, seq, dtype=torch.bool))
    for li in range(LAY):
        p = f"h.{li}."
        x = ln(hid, p + "ln_1.weight", p + "ln_1.bias")
        qkv = x @ t(p + "attn.c_attn.weight") + t(p + "attn.c_attn.bias")
        q, k, v = qkv.split(Hh, dim=-1)
        q = q.view(seq, HD, hd)
        k = k.view(seq, HD, hd)
        v = v.view(seq, HD, hd)
        att = torch.zeros(seq, HD, hd)
        for hh in range(HD):
            sc = (q[:, hh] @ k[:, hh].t()) / np.sqrt(hd)
            sc = sc.masked_fill(~mask, float("-inf"))
            att[:, hh] = torch.softmax(sc, -1) @ v[:, hh]
        hid = hid + att.reshape(seq, Hh) @ t(p + "attn.c_proj.weight") + \
            t(p + "attn.c_proj.bias")
        x = ln(hid, p + "ln_2.weight", p + "ln_2.bias")
        ff = gelu_new(x @ t(p + "mlp.c_fc.weight") + t(p + "mlp.c_fc.bias"))
        hid = hid + ff @ t(p + "mlp.c_proj.weight") + \
            t(p + "mlp.c_proj.bias")
    ref = ln(hid, "ln_f.weight", "ln_f.bias").numpy()
    err = np.abs(out - ref).max() / max(np.abs(ref).max(), 1e-6)
    assert err < 2e-3, err


def test_safetensors_sharded_checkpoint(tmp_path):
    """Sharded checkpoints (multiple *.safetensors in a dir) merge into
    one state dict and load identically to the single-file form."""
    from safetensors.numpy import load_file, save_file

    st = _mk_checkpoint(str(tmp_path))
    # re-save as two shards + remove the single file
    keys = sorted(st)
    half = len(keys) // 2
    save_file({k: st[k] for k in keys[:half]},
              os.path.join(str(tmp_path), "model-00001-of-00002"
                           ".safetensors"))
    save_file({k: st[k] for k in keys[half:]},
              os.path.join(str(tmp_path), "model-00002-of-00002"
                           ".safetensors"))
    single = os.path.join(str(tmp_path), "model.safetensors")
    ref_state = load_file(single)
    os.remove(single)

    g = build_llama_from_safetensors(str(tmp_path), batch=1, seq=16)
    plan = Planner().compile(g)
    ids = np.random.RandomState(2).randint(0, VOCAB, 16).astype(np.int32)
    out = run_reference(plan, ids)
    ref = _hf_forward(ref_state, torch.from_numpy(ids).long(), 16)
    assert np.abs(out - ref).max() / max(np.abs(ref).max(), 1e-6) < 2e-3
