"""Vision Transformer (ViT) builder — patch-embedding conv + learned
position constants + pre-LN transformer encoder + mean-pooled classifier
head, entirely on the engine's existing op set (conv2d / view / constant
/ add / layernorm / gemm+gelu / bidirectional attention / gavgpool).

Beyond-reference model family: the CUDA reference served CNN and BERT
TensorRT engines; ViT exercises the conv front-end AND the encoder
attention path in ONE graph. Mean pooling replaces the class token (the
common "gap" ViT variant) so the token count stays the patch grid
(image/patch)^2 — e.g. 196 for 224/16, an arbitrary non-multiple-of-64
sequence length the attention kernels handle via clamped tails.
"""
from __future__ import annotations

import numpy as np

from trtlab_amd.engine.ir import Graph


def build_vit(batch: int = 8, image: int = 224, patch: int = 16,
              hidden: int = 768, layers: int = 12, heads: int = 12,
              classes: int = 1000, seed: int = 0) -> Graph:
    """ViT-B/16 defaults. head_dim = hidden/heads must be 64 or 128."""
    assert image % patch == 0, "image must be a multiple of patch"
    hd = hidden // heads
    assert hidden % heads == 0 and hd in (64, 128), \
        "attention kernels: head_dim must be 64 or 128"
    grid = image // patch
    seq = grid * grid
    inter = hidden * 4
    rng = np.random.RandomState(seed)

    def w(nout, nin):
        return (rng.randn(nout, nin) * np.sqrt(1.0 / nin)).astype(np.float32)

    def b(n):
        return (rng.randn(n) * 0.02).astype(np.float32)

    def ln(n):
        return (rng.uniform(0.9, 1.1, n).astype(np.float32),
                (rng.randn(n) * 0.02).astype(np.float32))

    g = Graph(f"vit_p{patch}_h{hidden}_l{layers}_b{batch}")
    x = g.input((batch, image, image, 3), name="input")
    # patch embedding: conv k=patch s=patch -> NHWC [B, grid, grid, hidden];
    # NHWC is already token-major, so the token matrix is a pure view
    pw = (rng.randn(hidden, 3, patch, patch) *
          np.sqrt(1.0 / (3 * patch * patch))).astype(np.float32)
    h4 = g.conv2d(x, pw, stride=patch, padding=0, name="patch_embed")
    h = g.view(h4, (batch * seq, hidden), name="tokens")
    # learned position embedding, tiled across the batch as a device
    # constant and added to the tokens
    pos = (rng.randn(seq, hidden) * 0.02).astype(np.float32)
    pe = g.constant(np.tile(pos, (batch, 1)), name="pos_embed")
    h = g.add(h, pe, name="add_pos")
    for li in range(layers):
        g1, b1 = ln(hidden)
        xn = g.layernorm(h, g1, b1, name=f"l{li}_ln1")
        qkv = g.gemm(xn, w(3 * hidden, hidden), b(3 * hidden),
                     name=f"l{li}_qkv")
        att = g.attention(qkv, heads=heads, seq=seq, causal=False,
                          name=f"l{li}_att")
        proj = g.gemm(att, w(hidden, hidden), b(hidden),
                      name=f"l{li}_proj")
        h = g.add(h, proj, name=f"l{li}_res1")
        g2, b2 = ln(hidden)
        xn = g.layernorm(h, g2, b2, name=f"l{li}_ln2")
        ff1 = g.gemm(xn, w(inter, hidden), b(inter), name=f"l{li}_ff1")
        ff1 = g.gelu(ff1, name=f"l{li}_gelu")
        ff2 = g.gemm(ff1, w(hidden, inter), b(hidden), name=f"l{li}_ff2")
        h = g.add(h, ff2, name=f"l{li}_res2")
    gf, bf = ln(hidden)
    h = g.layernorm(h, gf, bf, name="ln_f")
    # mean-pool the patch grid per image -> [B, hidden] -> classifier
    hp = g.view(h, (batch, grid, grid, hidden), name="pool_in")
    pooled = g.global_avgpool(hp, name="pool")
    g.gemm(pooled, w(classes, hidden), b(classes), name="head")
    return g
