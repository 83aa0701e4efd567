#!/usr/bin/env python3
"""Generation-service load probe: N concurrent client streams against a
LLaMA DecodeSession behind the streaming RPC, measuring aggregate tok/s
(continuous batching: all live streams advance on every engine step).

    python tools/gen_load.py --batch 8 --clients 16 --new-tokens 64
"""
import argparse
import os
import sys
import time
from concurrent.futures import ThreadPoolExecutor

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import grpc  # noqa: E402
import numpy as np  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--clients", type=int, default=16)
    ap.add_argument("--layers", type=int, default=8)
    ap.add_argument("--hidden", type=int, default=1024)
    ap.add_argument("--prompt-len", type=int, default=16)
    ap.add_argument("--new-tokens", type=int, default=64)
    ap.add_argument("--inline-step", action="store_true")
    args = ap.parse_args()

    from trtlab_amd.engine.decode import DecodeSession
    from trtlab_amd.models import build_llama
    from trtlab_amd.rpc.generation import (GenerateRequest, GenerateToken,
                                           GenerationService)
    from trtlab_amd.rpc.server import Server

    g = build_llama(batch=args.batch, seq=1024, hidden=args.hidden,
                    layers=args.layers, heads=args.hidden // 128, seed=0)
    sess = DecodeSession(g, batch=args.batch, smax=1024, lm_head=True)
    svc = GenerationService(sess, inline_step=args.inline_step)
    srv = Server("127.0.0.1:0")
    srv.register_service(svc.service)
    srv.async_start()
    rng = np.random.RandomState(0)

    def one(i):
        ch = grpc.insecure_channel(f"127.0.0.1:{srv.port}")
        call = ch.stream_stream(
            "/trtlab.gen.Generation/Generate",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=GenerateToken.FromString)
        prompt = rng.randint(1, 30000, args.prompt_len).tolist()
        n = 0
        for r in call(iter([GenerateRequest(prompt=prompt,
                                            max_tokens=args.new_tokens)])):
            if not r.done:
                n += 1
        ch.close()
        return n

    t0 = time.perf_counter()
    with ThreadPoolExecutor(args.clients) as ex:
        counts = list(ex.map(one, range(args.clients)))
    dt = time.perf_counter() - t0
    total = sum(counts)
    print(f"{args.clients} streams x {args.new_tokens} tokens over "
          f"{args.batch} slots: {total} tokens in {dt:.2f}s = "
          f"{total / dt:,.0f} tok/s aggregate "
          f"({svc.engine.steps} engine steps)")
    srv.shutdown()


if __name__ == "__main__":
    main()
