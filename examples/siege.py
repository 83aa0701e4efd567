#!/usr/bin/env python3
"""Constant-rate load generator (reference: 02_TensorRT_GRPC/src/siege.cc:
constant-rate issue loop, max 950 outstanding)."""
import argparse
import gc
import sys
from pathlib import Path

import numpy as np

# constant-rate load generator: GC pauses show up directly as request-
# latency tail spikes — collect once up front, then disable (the issue
# loop allocates no reference cycles)
gc.disable()

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from trtlab_amd.rpc import InferRequest, InferResponse, siege


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--target", default="127.0.0.1:50051")
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--rate", type=float, default=100.0)
    ap.add_argument("--seconds", type=float, default=5.0)
    ap.add_argument("--max-outstanding", type=int, default=950)
    ap.add_argument("--shape", default="8,224,224,3")
    args = ap.parse_args()

    shape = tuple(int(s) for s in args.shape.split(","))
    batch = (np.random.RandomState(0).randn(*shape) * 0.5).astype(np.float16)
    # build the payload ONCE: re-serializing 1.2 MB of protobuf per request
    # makes the CLIENT the bottleneck at high rates (the server is 3
    # SO_REUSEPORT workers) — the load generator should generate load,
    # not burn its core on repeated identical serialization
    payload = batch.tobytes()
    req = InferRequest(model=args.model, input=payload, shape=list(shape),
                      dtype="f16")

    def make_request(i):
        return req

    stats = siege(args.target, "trtlab.Inference", "Compute", make_request,
                  InferResponse, rate_hz=args.rate, duration_s=args.seconds,
                  max_outstanding=args.max_outstanding)
    for k, v in stats.items():
        print(f"{k}: {v}")


if __name__ == "__main__":
    main()
