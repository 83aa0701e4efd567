#!/usr/bin/env python3
"""Continuous-batching LLM serving demo (GPU): a lockstep DecodeSession
whose slots independently start, finish, idle and restart requests —
with paged KV so parked slots hold zero cache memory.

Shows the full round-2 serving machinery:
  - per-slot device position counters (one replayed graph, no re-capture)
  - idle-slot masking (pos = -1 -> every per-slot kernel early-exits)
  - paged KV block tables (pages map on demand, recycle on completion)

    python examples/continuous_batching.py --batch 8 --requests 24
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=8, help="slots in lockstep")
    ap.add_argument("--requests", type=int, default=24)
    ap.add_argument("--layers", type=int, default=12)
    ap.add_argument("--max-new", type=int, default=48)
    ap.add_argument("--paged", action="store_true", default=True)
    args = ap.parse_args()

    from trtlab_amd.engine.decode import DecodeSession
    from trtlab_amd.models import build_gpt2

    rng = np.random.RandomState(0)
    g = build_gpt2(batch=1, seq=512, layers=args.layers, seed=0,
                   embeddings=True)
    sess = DecodeSession(g, batch=args.batch, smax=512, capture=True,
                         lm_head=True, paged=True)
    pool = sess.kv_pool

    # request queue: each request = (prompt token, length to generate)
    queue = [(int(rng.randint(1, 50000)), int(rng.randint(8, args.max_new)))
             for _ in range(args.requests)]
    slot_req = [-1] * args.batch       # which request a slot is serving
    slot_left = [0] * args.batch
    done = 0
    issued = 0
    ids = np.zeros(args.batch, np.int32)
    steps = 0

    # park everything, then admit work as slots free up
    for b in range(args.batch):
        sess.idle_slot(b)

    while done < args.requests:
        for b in range(args.batch):
            if slot_req[b] < 0 and issued < args.requests:
                tok, length = queue[issued]
                slot_req[b] = issued
                slot_left[b] = length
                issued += 1
                sess.reset_slot(b)     # fresh sequence, pages re-mapped
                ids[b] = tok
        logits = sess.step(ids)
        steps += 1
        nxt = logits.argmax(-1).astype(np.int32)
        for b in range(args.batch):
            if slot_req[b] < 0:
                continue
            slot_left[b] -= 1
            ids[b] = nxt[b]
            if slot_left[b] <= 0:
                done += 1
                slot_req[b] = -1
                sess.idle_slot(b)      # pages back to the pool immediately
        if steps % 16 == 0:
            active = sum(1 for r in slot_req if r >= 0)
            print(f"step {steps:4d}: {done}/{args.requests} done, "
                  f"{active} active slots, {pool.pages_free} pages free",
                  flush=True)

    print(f"served {done} requests in {steps} lockstep steps "
          f"({args.batch} slots); pages free at end: {pool.pages_free}")
    sess.close()


if __name__ == "__main__":
    main()
