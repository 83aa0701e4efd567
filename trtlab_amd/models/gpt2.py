"""GPT-2-style decoder IR builder (pre-LN transformer, causal attention),
random weights.

Beyond-reference model family: the reference served TensorRT encoder
engines only; this exercises the causal mask + online-softmax attention
path (csrc/kernels/attention.hip) at sequence lengths up to 1024. The
forward is a full-sequence pass (prefill); incremental KV-cache decode is
round-2 work.

Pre-LN block (GPT-2):
    h = h + proj(attn(qkv(ln1(h))))
    h = h + mlp(ln2(h))          # mlp = gemm -> gelu -> gemm
"""
from __future__ import annotations

import numpy as np

from trtlab_amd.engine.ir import Graph


def build_gpt2(batch: int = 8, seq: int = 1024, hidden: int = 768,
               layers: int = 12, heads: int = 12, seed: int = 0,
               embeddings: bool = False, vocab: int = 50257) -> Graph:
    """embeddings=True: int32 token ids input [B*S] -> tok+pos gather.
    Otherwise pre-embedded hidden states [B*S, hidden] fp16."""
    assert seq >= 1  # attention streams key tiles at ANY sequence length
    assert hidden % heads == 0 and hidden // heads in (64, 128), \
        "attention kernels: head_dim must be 64 or 128"
    inter = hidden * 4
    rng = np.random.RandomState(seed)

    def w(nout, nin):
        return (rng.randn(nout, nin) * np.sqrt(1.0 / nin)).astype(np.float32)

    def b(n):
        return (rng.randn(n) * 0.02).astype(np.float32)

    def ln(n):
        return (rng.uniform(0.9, 1.1, n).astype(np.float32),
                (rng.randn(n) * 0.02).astype(np.float32))

    g = Graph(f"gpt2_s{seq}_b{batch}")
    m = batch * seq
    if embeddings:
        ids = g.input((m,), name="token_ids", dtype="i32")
        tok = (rng.randn(vocab, hidden) * 0.02).astype(np.float32)
        pos = (rng.randn(seq, hidden) * 0.02).astype(np.float32)
        h = g.embedding(ids, tok, pos, name="embed")
    else:
        h = g.input((m, hidden), name="hidden_in")

    for li in range(layers):
        g1, b1 = ln(hidden)
        x = g.layernorm(h, g1, b1, name=f"l{li}_ln1")
        qkv = g.gemm(x, w(3 * hidden, hidden), b(3 * hidden),
                     name=f"l{li}_qkv")
        att = g.attention(qkv, heads=heads, seq=seq, causal=True,
                          name=f"l{li}_att")
        proj = g.gemm(att, w(hidden, hidden), b(hidden), name=f"l{li}_proj")
        h = g.add(h, proj, name=f"l{li}_res1")
        g2, b2 = ln(hidden)
        x = g.layernorm(h, g2, b2, name=f"l{li}_ln2")
        ff1 = g.gemm(x, w(inter, hidden), b(inter), name=f"l{li}_ff1")
        ff1 = g.gelu(ff1, name=f"l{li}_gelu")
        ff2 = g.gemm(ff1, w(hidden, inter), b(hidden),
                     name=f"l{li}_ff2")
        h = g.add(h, ff2, name=f"l{li}_res2")
    gf, bf = ln(hidden)
    g.layernorm(h, gf, bf, name="ln_f")
    return g
