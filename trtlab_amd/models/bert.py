"""BERT-base encoder IR builder (seq=128, fp16), random weights.

BASELINE config 5: "BERT-base seq=128 fp16 from ONNX (attention/GEMM MFMA
path, hipGraph-captured forward)". The graph takes pre-embedded hidden
states [B*S, hidden] as input (embedding gather is planned; the encoder
stack — QKV/attention/projection/FFN/LayerNorm — is the compute path the
baseline measures).
"""
from __future__ import annotations

import numpy as np

from trtlab_amd.engine.ir import Graph


def build_bert(batch: int = 8, seq: int = 128, hidden: int = 768,
               layers: int = 12, heads: int = 12, seed: int = 0,
               intermediate: int | None = None, embeddings: bool = False,
               varlen: bool = False, pad_id: int = 0,
               vocab: int = 30522, segments: bool = False,
               mask_input: bool = False) -> Graph:
    """embeddings=True: input is int32 token ids [B*S]; the graph starts
    with an embedding gather (tok+pos) + LayerNorm. Otherwise the input is
    the pre-embedded hidden state [B*S, hidden] fp16.
    varlen=True (needs embeddings): right-padded variable-length batches —
    per-sequence valid lengths are derived on-device from ids != pad_id and
    padded keys are masked out of every attention softmax.
    segments=True: adds a segment_ids i32 input binding + learned
    token-type table (BERT token_type_ids). mask_input=True: adds an
    attention_mask i32 input binding that drives the varlen key masking
    instead of the pad-id scan — together these express BERT with its REAL
    (ids, mask, segments) bindings (reference bindings.h:60-120 carves one
    device address per binding the same way)."""
    if varlen and not embeddings:
        raise ValueError("varlen requires embeddings=True (ids input)")
    if (segments or mask_input) and not embeddings:
        raise ValueError("segments/mask_input require embeddings=True")
    if mask_input and not varlen:
        raise ValueError("mask_input requires varlen=True")
    inter = intermediate or hidden * 4
    rng = np.random.RandomState(seed)

    def w(nout, nin, scale=None):
        s = scale or np.sqrt(1.0 / nin)
        return (rng.randn(nout, nin) * s).astype(np.float32)

    def b(n, scale=0.02):
        return (rng.randn(n) * scale).astype(np.float32)

    def ln(n):
        return (rng.uniform(0.9, 1.1, n).astype(np.float32),
                (rng.randn(n) * 0.02).astype(np.float32))

    g = Graph(f"bert_base_s{seq}_b{batch}")
    m = batch * seq
    if embeddings:
        ids = g.input((m,), name="token_ids", dtype="i32")
        segids = (g.input((m,), name="segment_ids", dtype="i32")
                  if segments else None)
        if mask_input:
            g.input((m,), name="attention_mask", dtype="i32")
        tok = (rng.randn(vocab, hidden) * 0.02).astype(np.float32)
        pos = (rng.randn(seq, hidden) * 0.02).astype(np.float32)
        seg = ((rng.randn(2, hidden) * 0.02).astype(np.float32)
               if segments else None)
        emb = g.embedding(ids, tok, pos, seg_table=seg, segids=segids,
                          name="embed")
        ge, be = ln(hidden)
        h = g.layernorm(emb, ge, be, name="embed_ln")
    else:
        h = g.input((m, hidden), name="hidden_in")
    for li in range(layers):
        qkv = g.gemm(h, w(3 * hidden, hidden), b(3 * hidden),
                     name=f"l{li}_qkv")
        att = g.attention(qkv, heads=heads, seq=seq, varlen=varlen,
                          pad_id=pad_id, name=f"l{li}_att")
        proj = g.gemm(att, w(hidden, hidden), b(hidden), name=f"l{li}_proj")
        ga, ba = ln(hidden)
        h1 = g.add_layernorm(proj, h, ga, ba, name=f"l{li}_ln1")
        ff1 = g.gemm(h1, w(inter, hidden), b(inter), name=f"l{li}_ff1")
        ff1 = g.gelu(ff1, name=f"l{li}_gelu")
        ff2 = g.gemm(ff1, w(hidden, inter), b(hidden), name=f"l{li}_ff2")
        gb, bb = ln(hidden)
        h = g.add_layernorm(ff2, h1, gb, bb, name=f"l{li}_ln2")
    return g
