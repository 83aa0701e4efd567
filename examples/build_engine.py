#!/usr/bin/env python3
"""Engine builder CLI (reference: examples/ONNX/resnet50/build.py — ONNX ->
serialized engine). Compiles an .onnx (or a builtin model) to a plan cache:

  python examples/build_engine.py --onnx model.onnx --batch 8 \
      --dtype int8 --out model_b8_int8.npz
  python examples/build_engine.py --model resnet50 --out rn50.npz
"""
import argparse
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--onnx")
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--dtype", default="fp16", choices=["fp16", "int8", "fp8"])
    ap.add_argument("--out", required=True)
    args = ap.parse_args()

    from trtlab_amd.engine.plan_io import save_plan
    from trtlab_amd.engine.planner import DT_F16, DT_F8, DT_I8, Planner
    from trtlab_amd.models import build_bert, build_resnet

    t0 = time.time()
    if args.onnx:
        from trtlab_amd.engine.onnx_io import load_onnx

        g = load_onnx(args.onnx, batch=args.batch)
    elif args.model == "bert":
        g = build_bert(batch=args.batch, seq=128, layers=12, seed=0)
    else:
        g = build_resnet(int(args.model.replace("resnet", "")),
                         batch=args.batch, seed=0)
    dtype = {"fp16": DT_F16, "int8": DT_I8, "fp8": DT_F8}[args.dtype]
    plan = Planner(dtype=dtype).compile(g)
    save_plan(plan, args.out)
    print(f"built {plan.name}: {len(plan.ops)} ops, "
          f"weights {plan.weights.nbytes >> 20} MiB, "
          f"arena {plan.arena_bytes >> 20} MiB -> {args.out} "
          f"({time.time() - t0:.1f}s)")


if __name__ == "__main__":
    main()
