#!/usr/bin/env python3
"""Inference clients (reference: 02_TensorRT_GRPC client-sync.x /
client-async.x). Sends synthetic batches, prints throughput + latency."""
import argparse
import sys
import time
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from trtlab_amd.rpc import AsyncClient, InferRequest, InferResponse, SyncClient


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--target", default="127.0.0.1:50051")
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--count", type=int, default=100)
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--shape", default="8,224,224,3")
    ap.add_argument("--mode", choices=["sync", "async"], default="sync")
    args = ap.parse_args()

    shape = tuple(int(s) for s in args.shape.split(","))
    batch = (np.random.RandomState(0).randn(*shape) * 0.5).astype(np.float16)
    req = InferRequest(model=args.model, input=batch.tobytes(),
                       shape=list(shape), dtype="f16")

    lat = []
    t0 = time.monotonic()
    if args.mode == "sync":
        c = SyncClient(args.target)
        for i in range(args.count):
            t = time.monotonic()
            resp = c.call("trtlab.Inference", "Compute", req, InferResponse,
                          timeout=60)
            lat.append(time.monotonic() - t)
        c.close()
    else:
        c = AsyncClient(args.target)
        futs = [c.call("trtlab.Inference", "Compute", req, InferResponse,
                       timeout=120) for _ in range(args.count)]
        for f in futs:
            f.result(timeout=120)
        c.close()
    elapsed = time.monotonic() - t0
    inf = args.count * shape[0]
    print(f"{args.count} requests ({inf} inferences) in {elapsed:.3f}s "
          f"-> {inf/elapsed:.1f} inf/sec")
    if lat:
        lat_ms = np.array(sorted(lat)) * 1e3
        print(f"latency p50 {np.percentile(lat_ms, 50):.2f} ms / "
              f"p90 {np.percentile(lat_ms, 90):.2f} / "
              f"p99 {np.percentile(lat_ms, 99):.2f}")


if __name__ == "__main__":
    main()
