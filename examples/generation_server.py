#!/usr/bin/env python3
"""Streaming LLM generation service (GPU): continuous batching over an
incremental DecodeSession behind a single-up / token-stream-down gRPC
surface (reference streaming lifecycle shapes; the decode engine is
beyond-reference).

    python examples/generation_server.py --arch llama --port 50055
    python examples/generation_server.py --client --port 50055 \
        --prompt 11 42 7 --new-tokens 32
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def serve(args):
    from trtlab_amd.engine.decode import DecodeSession
    from trtlab_amd.models import (build_gpt2, build_llama,
                                   build_llama_from_safetensors)
    from trtlab_amd.rpc.generation import GenerationService
    from trtlab_amd.rpc.server import Server

    if args.ckpt:
        # real HF weights (single file or sharded dir), arch-selected
        from trtlab_amd.models import build_gpt2_from_safetensors

        loader = (build_llama_from_safetensors if args.arch == "llama"
                  else build_gpt2_from_safetensors)
        g = loader(args.ckpt, batch=args.batch, seq=args.smax)
    elif args.arch == "llama":
        hid = args.hidden or 1024
        g = build_llama(batch=args.batch, seq=args.smax, hidden=hid,
                        layers=args.layers, heads=hid // 128, seed=0)
    else:
        g = build_gpt2(batch=args.batch, seq=args.smax,
                       hidden=args.hidden or 768, layers=args.layers,
                       seed=0, embeddings=True)
    sess = DecodeSession(g, batch=args.batch, smax=args.smax, lm_head=True)
    svc = GenerationService(sess)
    srv = Server(f"0.0.0.0:{args.port}")
    srv.register_service(svc.service)
    srv.async_start()
    print(f"generation server ({args.arch}, batch {args.batch}) "
          f"on :{srv.port}; ctrl-c to stop", flush=True)
    try:
        while True:
            time.sleep(2)
    except KeyboardInterrupt:
        srv.shutdown()


def client(args):
    import grpc

    from trtlab_amd.rpc.generation import GenerateRequest, GenerateToken

    ch = grpc.insecure_channel(f"127.0.0.1:{args.port}")
    call = ch.stream_stream(
        "/trtlab.gen.Generation/Generate",
        request_serializer=lambda m: m.SerializeToString(),
        response_deserializer=GenerateToken.FromString)
    t0 = time.perf_counter()
    toks = []
    req = GenerateRequest(prompt=args.prompt, max_tokens=args.new_tokens,
                          temperature=args.temperature, top_k=args.top_k,
                          top_p=args.top_p, seed=args.seed)
    for resp in call(iter([req])):
        if not resp.done:
            toks.append(resp.token)
            print(f"  token[{resp.index}] = {resp.token} "
                  f"(slot {resp.slot})", flush=True)
    dt = time.perf_counter() - t0
    print(f"{len(toks)} tokens in {dt:.2f}s "
          f"({len(toks) / max(dt, 1e-9):,.0f} tok/s single stream)")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--client", action="store_true")
    ap.add_argument("--arch", choices=("gpt2", "llama"), default="gpt2")
    ap.add_argument("--ckpt", default="",
                    help="HF LLaMA safetensors checkpoint (file or dir)")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--layers", type=int, default=8)
    ap.add_argument("--hidden", type=int, default=0)
    ap.add_argument("--smax", type=int, default=1024)
    ap.add_argument("--port", type=int, default=50055)
    ap.add_argument("--prompt", type=int, nargs="*", default=[11, 42, 7])
    ap.add_argument("--new-tokens", type=int, default=32)
    ap.add_argument("--temperature", type=float, default=0.0)
    ap.add_argument("--top-k", type=int, default=0)
    ap.add_argument("--top-p", type=float, default=0.0)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    client(args) if args.client else serve(args)


if __name__ == "__main__":
    main()
