#!/usr/bin/env python3
"""Streaming inference service (reference: nvrpc StreamingService example +
life_cycle_streaming.h): a bidirectional stream of single-image requests;
responses stream back in order as the windowed batcher fills engine batches.
Demonstrates the streaming lifecycle + cyclic windowed buffering on the
inference path.

  python examples/streaming_server.py --model resnet50 --port 50053
"""
import argparse
import asyncio
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--port", type=int, default=50053)
    ap.add_argument("--contexts", type=int, default=2)
    args = ap.parse_args()

    from trtlab_amd.engine.planner import Planner
    from trtlab_amd.engine.runtime import InferenceManager
    from trtlab_amd.models import build_resnet
    from trtlab_amd.rpc import InferRequest, InferResponse
    from trtlab_amd.rpc.server import Server, StreamingService

    depth = int(args.model.replace("resnet", ""))
    g = build_resnet(depth, batch=args.batch, seed=0)
    plan = Planner().compile(g)
    mgr = InferenceManager(max_contexts=args.contexts)
    mgr.register_model(args.model, plan)
    mgr.allocate_resources()
    runner = mgr.infer_runner(args.model)
    ishape = plan.input_shape

    svc = StreamingService("trtlab.Inference")

    async def stream_compute(request_iter, context, resources):
        """Collect up to `batch` in-flight requests, run, stream back."""
        pending = []

        async def flush():
            batch = np.zeros(ishape, dtype=np.float16)
            for i, r in enumerate(pending):
                batch[i] = np.frombuffer(
                    r.input, dtype=np.float16).reshape(ishape[1:])
            out = await asyncio.wrap_future(runner.infer(batch))
            for i, r in enumerate(pending):
                yield InferResponse(output=out[i].tobytes(),
                                    shape=list(out[i].shape), dtype="f16",
                                    batch_id=r.batch_id)
            pending.clear()

        async for req in request_iter:
            pending.append(req)
            if len(pending) >= args.batch:
                async for resp in flush():
                    yield resp
        if pending:
            async for resp in flush():
                yield resp

    svc.register_streaming("ComputeStream", stream_compute, InferRequest,
                           InferResponse)
    server = Server(f"0.0.0.0:{args.port}")
    server.register_service(svc)
    print(f"streaming server on :{args.port}")
    server.run()


if __name__ == "__main__":
    main()
