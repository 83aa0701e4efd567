#!/usr/bin/env python3
"""Local pipelined throughput benchmark via InferBench (reference:
examples/00_TensorRT infer.x / inference.x — saturate the context pool for
N seconds, report batches/sec, inf/sec, latency quantiles)."""
import argparse
import sys
from pathlib import Path

import numpy as np

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--contexts", type=int, default=3)
    ap.add_argument("--seconds", type=float, default=5.0)
    ap.add_argument("--dtype", default="fp16", choices=["fp16", "int8"])
    ap.add_argument("--onnx", default=None, help="load model from .onnx")
    ap.add_argument("--plan", default=None, help="load a saved .npz plan")
    args = ap.parse_args()

    from trtlab_amd.engine.planner import DT_F16, DT_I8, Planner
    from trtlab_amd.engine.runtime import InferBench, InferenceManager
    from trtlab_amd.models import build_bert, build_resnet

    mgr = InferenceManager(max_contexts=args.contexts)
    if args.plan:
        mgr.register_plan_file(args.model, args.plan)
    elif args.onnx:
        mgr.register_onnx(args.model, args.onnx, batch=args.batch,
                          dtype=DT_I8 if args.dtype == "int8" else DT_F16)
    else:
        if args.model == "bert":
            g = build_bert(batch=args.batch, seq=128, layers=12, seed=0)
        else:
            g = build_resnet(int(args.model.replace("resnet", "")),
                             batch=args.batch, seed=0)
        plan = Planner(dtype=DT_I8 if args.dtype == "int8" else DT_F16).compile(g)
        mgr.register_model(args.model, plan)
    mgr.allocate_resources()

    plan = mgr.get_model(args.model).plan
    x = (np.random.RandomState(0).randn(*plan.input_shape) * 0.5).astype(np.float16)
    bench = InferBench(mgr, args.model)
    stats = bench.run(x, seconds=args.seconds,
                      max_outstanding=args.contexts * 2)
    for k, v in stats.items():
        print(f"{k}: {v}")
    mgr.shutdown()


if __name__ == "__main__":
    main()
