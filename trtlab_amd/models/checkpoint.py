"""Checkpoint interop: build engine graphs from HuggingFace-format
safetensors weights (LLaMA family). The reference deserialized TensorRT
`.engine` blobs (trtlab/tensorrt/runtime.h:43); here the portable
interchange is the HF safetensors layout, mapped onto the same IR the
random-init builders produce — so a user with real LLaMA weights gets
the full planner/kernel/serving stack unchanged.

Weight-name mapping (HF `LlamaForCausalLM`):
    model.embed_tokens.weight                   -> token table [V, H]
    model.layers.{i}.input_layernorm.weight     -> rms1 gamma
    model.layers.{i}.self_attn.{q,k,v}_proj     -> fused qkv [3H, H]
    model.layers.{i}.self_attn.o_proj           -> proj
    model.layers.{i}.post_attention_layernorm   -> rms2 gamma
    model.layers.{i}.mlp.{gate,up,down}_proj    -> SwiGLU
    model.norm.weight                           -> final rmsnorm

nn.Linear stores [out, in] and computes x @ W^T — exactly the engine's
gemm weight contract. HF rotary = half-split rotate (rotate_half) with
angle base `rope_theta`, the same convention as csrc rope_kernel.
GQA checkpoints (num_key_value_heads < num_attention_heads) are
loaded by replicating each kv head's k/v projection rows across its
query group — numerically IDENTICAL to grouped attention (every query
head in group g attends the same k/v), traded for MHA-sized KV cache
(a fine trade against 288 GB of HBM3E per GPU).
"""
from __future__ import annotations

import json
import os
from typing import Dict, Optional

import numpy as np

from trtlab_amd.engine.ir import Graph


def _load_state(path: str) -> Dict[str, np.ndarray]:
    """Read one .safetensors file or every *.safetensors in a directory
    (sharded checkpoints) into float32 numpy arrays."""
    from safetensors.numpy import load_file

    files = []
    if os.path.isdir(path):
        files = sorted(
            os.path.join(path, f) for f in os.listdir(path)
            if f.endswith(".safetensors"))
        if not files:
            raise FileNotFoundError(f"no *.safetensors under {path}")
    else:
        files = [path]
    state: Dict[str, np.ndarray] = {}
    for f in files:
        for k, v in load_file(f).items():
            state[k] = np.asarray(v, np.float32)
    return state


def _read_config(path: str) -> dict:
    cfg = {}
    d = path if os.path.isdir(path) else os.path.dirname(path)
    cj = os.path.join(d, "config.json")
    if os.path.exists(cj):
        with open(cj) as f:
            cfg = json.load(f)
    return cfg


def build_llama_from_safetensors(path: str, batch: int = 1,
                                 seq: int = 256,
                                 heads: Optional[int] = None,
                                 theta: Optional[float] = None) -> Graph:
    """Load HF LLaMA weights and build the engine IR graph around them.
    heads/theta default from config.json next to the checkpoint."""
    state = _load_state(path)
    cfg = _read_config(path)
    heads = heads or cfg.get("num_attention_heads")
    if heads is None:
        raise ValueError("pass heads= (no config.json found)")
    kvh = int(cfg.get("num_key_value_heads", heads))
    if heads % kvh != 0:
        raise ValueError(f"heads {heads} not divisible by kv heads {kvh}")
    theta = theta or float(cfg.get("rope_theta", 10000.0))
    eps = float(cfg.get("rms_norm_eps", 1e-5))

    tok = state["model.embed_tokens.weight"]  # [V, H]
    vocab, hidden = tok.shape
    hd = hidden // heads
    if hd not in (64, 128):
        raise ValueError(f"head_dim {hd} unsupported (attention kernels "
                         "need 64 or 128)")
    layers = 0
    while f"model.layers.{layers}.input_layernorm.weight" in state:
        layers += 1
    if layers == 0:
        raise ValueError("no model.layers.* in checkpoint")

    g = Graph(f"llama_hf_h{hidden}_l{layers}_s{seq}_b{batch}")
    m = batch * seq
    ids = g.input((m,), name="token_ids", dtype="i32")
    zpos = np.zeros((seq, hidden), np.float32)
    h = g.embedding(ids, tok, zpos, name="embed")
    for li in range(layers):
        pre = f"model.layers.{li}."
        x = g.rmsnorm(h, state[pre + "input_layernorm.weight"], eps=eps,
                      name=f"l{li}_rms1")
        def expand_kv(w):
            # GQA -> MHA: repeat each kv head's hd-row block across its
            # query group (exact — same k/v seen by every head in the
            # group, rope rotation depends only on the in-head dim)
            if kvh == heads:
                return w
            r = heads // kvh
            blocks = w.reshape(kvh, hd, w.shape[1])
            return np.repeat(blocks, r, axis=0).reshape(-1, w.shape[1])

        qkv_w = np.concatenate(
            [state[pre + "self_attn.q_proj.weight"],
             expand_kv(state[pre + "self_attn.k_proj.weight"]),
             expand_kv(state[pre + "self_attn.v_proj.weight"])], axis=0)
        qkv = g.gemm(x, qkv_w, None, name=f"l{li}_qkv")
        qkv = g.rope(qkv, heads=heads, seq=seq, theta=theta,
                     name=f"l{li}_rope")
        att = g.attention(qkv, heads=heads, seq=seq, causal=True,
                          name=f"l{li}_att")
        proj = g.gemm(att, state[pre + "self_attn.o_proj.weight"], None,
                      name=f"l{li}_proj")
        h = g.add(h, proj, name=f"l{li}_res1")
        x = g.rmsnorm(h, state[pre + "post_attention_layernorm.weight"],
                      eps=eps, name=f"l{li}_rms2")
        gate = g.gemm(x, state[pre + "mlp.gate_proj.weight"], None,
                      name=f"l{li}_gate")
        up = g.gemm(x, state[pre + "mlp.up_proj.weight"], None,
                    name=f"l{li}_up")
        ff = g.silu_mul(gate, up, name=f"l{li}_swiglu")
        down = g.gemm(ff, state[pre + "mlp.down_proj.weight"], None,
                      name=f"l{li}_down")
        h = g.add(h, down, name=f"l{li}_res2")
    g.rmsnorm(h, state["model.norm.weight"], eps=eps, name="rms_f")
    return g


def build_gpt2_from_safetensors(path: str, batch: int = 1,
                                seq: int = 256,
                                heads: Optional[int] = None) -> Graph:
    """Load HF GPT-2 weights (`GPT2LMHeadModel` safetensors) into the
    engine IR. HF GPT-2 uses Conv1D (x @ W, weight [in, out]) where the
    engine's gemm computes x @ W^T — every c_attn/c_proj/c_fc weight is
    transposed on load. Names may carry the `transformer.` prefix.

    Mapping: wte/wpe -> embedding tables; h.{i}.ln_1/ln_2 -> layernorms;
    h.{i}.attn.c_attn -> fused qkv; attn.c_proj -> proj; mlp.c_fc/c_proj
    -> ff1/ff2 (GELU between, HF gelu_new == the engine's tanh GELU);
    ln_f -> final layernorm."""
    state = _load_state(path)
    cfg = _read_config(path)
    pre = "transformer." if any(k.startswith("transformer.")
                                for k in state) else ""

    def get(name):
        return state[pre + name]

    heads = heads or cfg.get("n_head")
    if heads is None:
        raise ValueError("pass heads= (no config.json with n_head)")
    tok = get("wte.weight")                 # [V, H]
    posw = get("wpe.weight")                # [P, H]
    vocab, hidden = tok.shape
    if hidden // heads not in (64, 128):
        raise ValueError(f"head_dim {hidden // heads} unsupported")
    if seq > posw.shape[0]:
        raise ValueError(f"seq {seq} > position table {posw.shape[0]}")
    layers = 0
    while f"{pre}h.{layers}.ln_1.weight" in state:
        layers += 1
    if layers == 0:
        raise ValueError("no h.* blocks in checkpoint")

    g = Graph(f"gpt2_hf_h{hidden}_l{layers}_s{seq}_b{batch}")
    m = batch * seq
    ids = g.input((m,), name="token_ids", dtype="i32")
    h = g.embedding(ids, tok, posw[:seq], name="embed")
    for li in range(layers):
        p = f"h.{li}."
        x = g.layernorm(h, get(p + "ln_1.weight"), get(p + "ln_1.bias"),
                        name=f"l{li}_ln1")
        qkv = g.gemm(x, get(p + "attn.c_attn.weight").T.copy(),
                     get(p + "attn.c_attn.bias"), name=f"l{li}_qkv")
        att = g.attention(qkv, heads=heads, seq=seq, causal=True,
                          name=f"l{li}_att")
        proj = g.gemm(att, get(p + "attn.c_proj.weight").T.copy(),
                      get(p + "attn.c_proj.bias"), name=f"l{li}_proj")
        h = g.add(h, proj, name=f"l{li}_res1")
        x = g.layernorm(h, get(p + "ln_2.weight"), get(p + "ln_2.bias"),
                        name=f"l{li}_ln2")
        ff1 = g.gemm(x, get(p + "mlp.c_fc.weight").T.copy(),
                     get(p + "mlp.c_fc.bias"), name=f"l{li}_ff1")
        ff1 = g.gelu(ff1, name=f"l{li}_gelu")
        ff2 = g.gemm(ff1, get(p + "mlp.c_proj.weight").T.copy(),
                     get(p + "mlp.c_proj.bias"), name=f"l{li}_ff2")
        h = g.add(h, ff2, name=f"l{li}_res2")
    g.layernorm(h, get("ln_f.weight"), get("ln_f.bias"), name="ln_f")
    return g
