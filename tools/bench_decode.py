#!/usr/bin/env python3
"""GPT-2 incremental decode micro-benchmark (KV cache + hipGraph replay)."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from trtlab_amd.engine.decode import DecodeSession
from trtlab_amd.models import build_gpt2


def main():
    for batch in (8, 64):
        g = build_gpt2(batch=batch, seq=1024, layers=12, seed=0,
                       embeddings=True)
        sess = DecodeSession(g, batch=batch, smax=1024, capture=True)
        rng = np.random.RandomState(0)
        ids = rng.randint(1, 50257, (batch,)).astype(np.int32)
        for _ in range(20):  # warmup + capture
            sess.step(ids)
        n = 200
        t0 = time.perf_counter()
        for _ in range(n):
            sess.step(ids)
        dt = (time.perf_counter() - t0) / n
        print(f"b{batch}: {dt*1e6:7.1f} us/step = "
              f"{batch/dt:10.0f} tok/s  ({1e3*dt:.3f} ms/token-step)",
              flush=True)
        sess.close()


if __name__ == "__main__":
    main()
