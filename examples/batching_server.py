#!/usr/bin/env python3
"""Dynamic-batching inference server (reference: 03_Batching
inference-batcher + LifeCycleBatching; BASELINE config 3 serving shape).

Accepts single-image requests, collects them into engine-sized batches via
the core Dispatcher (max_batch_size / timeout window), pads short batches,
and runs the int8 (or fp16) ResNet-50 engine with N concurrent HIP-stream
contexts.

  python examples/batching_server.py --dtype int8 --contexts 4
"""
import argparse
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--port", type=int, default=50052)
    ap.add_argument("--contexts", type=int, default=4)
    ap.add_argument("--dtype", default="int8", choices=["fp16", "int8"])
    ap.add_argument("--window-ms", type=float, default=2.0)
    args = ap.parse_args()

    from trtlab_amd.engine.planner import DT_F16, DT_I8, Planner
    from trtlab_amd.engine.runtime import InferenceManager
    from trtlab_amd.models import build_resnet
    from trtlab_amd.rpc import InferRequest, InferResponse
    from trtlab_amd.rpc.server import BatchingService, Server

    depth = int(args.model.replace("resnet", ""))
    g = build_resnet(depth, batch=args.batch, seed=0)
    plan = Planner(dtype=DT_I8 if args.dtype == "int8" else DT_F16).compile(g)

    mgr = InferenceManager(max_contexts=args.contexts)
    mgr.register_model(args.model, plan)
    mgr.allocate_resources()
    runner = mgr.infer_runner(args.model)
    ishape = plan.input_shape

    def compute_batch(requests):
        """Pad to the engine batch, run once, scatter responses."""
        n = len(requests)
        batch = np.zeros(ishape, dtype=np.float16)
        for i, r in enumerate(requests):
            batch[i] = np.frombuffer(r.input, dtype=np.float16).reshape(ishape[1:])
        out = runner.infer(batch).result(timeout=60)
        return [InferResponse(output=out[i].tobytes(),
                              shape=list(out[i].shape), dtype="f16",
                              batch_id=r.batch_id)
                for i, r in enumerate(requests)]

    server = Server(f"0.0.0.0:{args.port}")
    svc = BatchingService("trtlab.Inference", "Compute", InferRequest,
                          InferResponse, compute_batch,
                          max_batch_size=args.batch,
                          timeout_s=args.window_ms / 1e3, workers=args.contexts)
    server.register_service(svc)
    print(f"batching server: {args.model} {args.dtype} window "
          f"{args.window_ms} ms batch {args.batch} on :{args.port}")
    server.run()


if __name__ == "__main__":
    main()
