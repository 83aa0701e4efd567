"""LLaMA-family decoder builder (pre-norm RMSNorm + RoPE + SwiGLU,
head_dim 128, no biases) — the modern-LLM architecture on top of the
engine's causal attention / rmsnorm / rope / silu_mul ops. Random
weights (no checkpoints offline), synthetic-scale initialized like the
other builders. Beyond-reference: the CUDA reference served static CNN/
BERT TensorRT engines only."""
from __future__ import annotations

import numpy as np

from trtlab_amd.engine.ir import Graph


def build_llama(batch: int = 1, seq: int = 256, hidden: int = 1024,
                layers: int = 4, heads: int = 8, seed: int = 0,
                vocab: int = 32000, theta: float = 10000.0,
                intermediate: int | None = None) -> Graph:
    """head_dim = hidden // heads must be 64 or 128 (attention kernels).
    Input: int32 token ids [B*S]; output: final RMS-normed hidden states
    [B*S, hidden] (weight-tied scoring is hidden @ tok^T, as in GPT-2)."""
    hd = hidden // heads
    assert hidden % heads == 0 and hd in (64, 128), "head_dim must be 64/128"
    inter = intermediate or int(round(hidden * 8 / 3 / 64) * 64)
    rng = np.random.RandomState(seed)

    def w(nout, nin):
        return (rng.randn(nout, nin) * np.sqrt(1.0 / nin)).astype(np.float32)

    g = Graph(f"llama_h{hidden}_l{layers}_s{seq}_b{batch}")
    m = batch * seq
    ids = g.input((m,), name="token_ids", dtype="i32")
    tok = (rng.randn(vocab, hidden) * 0.02).astype(np.float32)
    # RoPE replaces additive position embeddings: a zero pos table keeps
    # the embedding op's contract (out = tok[ids] + 0)
    zpos = np.zeros((seq, hidden), np.float32)
    h = g.embedding(ids, tok, zpos, name="embed")
    for li in range(layers):
        gam1 = (rng.uniform(0.9, 1.1, hidden)).astype(np.float32)
        x = g.rmsnorm(h, gam1, name=f"l{li}_rms1")
        qkv = g.gemm(x, w(3 * hidden, hidden), None, name=f"l{li}_qkv")
        qkv = g.rope(qkv, heads=heads, seq=seq, theta=theta,
                     name=f"l{li}_rope")
        att = g.attention(qkv, heads=heads, seq=seq, causal=True,
                          name=f"l{li}_att")
        proj = g.gemm(att, w(hidden, hidden), None, name=f"l{li}_proj")
        h = g.add(h, proj, name=f"l{li}_res1")
        gam2 = (rng.uniform(0.9, 1.1, hidden)).astype(np.float32)
        x = g.rmsnorm(h, gam2, name=f"l{li}_rms2")
        gate = g.gemm(x, w(inter, hidden), None, name=f"l{li}_gate")
        up = g.gemm(x, w(inter, hidden), None, name=f"l{li}_up")
        ff = g.silu_mul(gate, up, name=f"l{li}_swiglu")
        down = g.gemm(ff, w(hidden, inter), None, name=f"l{li}_down")
        h = g.add(h, down, name=f"l{li}_res2")
    gf = (rng.uniform(0.9, 1.1, hidden)).astype(np.float32)
    g.rmsnorm(h, gf, name="rms_f")
    return g
