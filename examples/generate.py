#!/usr/bin/env python3
"""Greedy generation on the incremental decode path (GPU) — GPT-2 or LLaMA.

Random-init weights (no network for checkpoints), so the "text" is noise —
this demonstrates the serving mechanics: one hipGraph replay per token
against resident KV caches, with logits from the weight-tied lm head.

    python examples/generate.py --batch 4 --prompt-len 16 --new-tokens 32
    python examples/generate.py --arch llama --hidden 1024 --layers 8
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from trtlab_amd.engine.decode import DecodeSession
from trtlab_amd.models import build_gpt2, build_llama


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=4)
    ap.add_argument("--arch", choices=("gpt2", "llama"), default="gpt2")
    ap.add_argument("--hidden", type=int, default=0,
                    help="hidden size (default: 768 gpt2 / 1024 llama)")
    ap.add_argument("--layers", type=int, default=12)
    ap.add_argument("--prompt-len", type=int, default=16)
    ap.add_argument("--new-tokens", type=int, default=32)
    ap.add_argument("--speculative", type=int, default=0, metavar="K",
                    help="speculative decoding with a draft of "
                         "--draft-layers layers proposing K tokens/round "
                         "(output provably identical to plain greedy)")
    ap.add_argument("--draft-layers", type=int, default=2)
    args = ap.parse_args()

    def build(layers):
        if args.arch == "llama":
            hid = args.hidden or 1024
            return build_llama(batch=args.batch, seq=1024, hidden=hid,
                               layers=layers, heads=hid // 128, seed=0)
        return build_gpt2(batch=args.batch, seq=1024, layers=layers,
                          hidden=args.hidden or 768, seed=0,
                          embeddings=True)

    g = build(args.layers)
    vocab = 32000 if args.arch == "llama" else 50257
    sess = DecodeSession(g, batch=args.batch, smax=1024, lm_head=True)

    rng = np.random.RandomState(0)
    prompt = rng.randint(1, vocab,
                         (args.batch, args.prompt_len)).astype(np.int32)
    logits = sess.prefill(prompt)  # one fused pass fills the KV caches

    toks = np.argmax(logits, axis=1).astype(np.int32)
    if args.speculative:
        # draft (same weights family, fewer layers) proposes K per round;
        # the target verifies each chunk in ONE forward
        from trtlab_amd.engine.decode import SpeculativeDecoder

        gd = build(args.draft_layers)
        # capture=True: verification chunks + draft steps replay as
        # hipGraphs (per chunk size) instead of eager per-kernel launches
        draft = DecodeSession(gd, batch=args.batch, smax=1024,
                              capture=True, lm_head=True)
        draft.prefill(prompt)
        target = DecodeSession(g, batch=args.batch, smax=1024,
                               capture=True, lm_head=True)
        target.prefill(prompt)
        sd = SpeculativeDecoder(target, draft, k=args.speculative)
        t0 = time.perf_counter()
        seqs, rate_acc = sd.generate(toks, args.new_tokens - 1)
        dt = time.perf_counter() - t0
        seqs = np.concatenate([toks[:, None], seqs], axis=1)
        print(f"speculative acceptance rate: {rate_acc:.2f}")
        target.close()
        draft.close()
    else:
        out = [toks]
        t0 = time.perf_counter()
        for _ in range(args.new_tokens - 1):
            # in-graph argmax head: B ints over PCIe instead of logits
            toks = sess.step(toks, return_ids=True).astype(np.int32)
            out.append(toks)
        dt = time.perf_counter() - t0
        seqs = np.stack(out, axis=1)
    print("generated token ids (greedy):")
    for b in range(args.batch):
        print(f"  seq{b}: {seqs[b][:16].tolist()} ...")
    rate = args.batch * (args.new_tokens - 1) / dt
    print(f"decode rate: {rate:,.0f} tok/s "
          f"({dt / (args.new_tokens - 1) * 1e3:.2f} ms/step, "
          f"batch {args.batch}, lm head incl.)")
    sess.close()


if __name__ == "__main__":
    main()
