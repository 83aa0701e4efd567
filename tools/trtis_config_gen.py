#!/usr/bin/env python3
"""Emit a TRTIS model_config.pbtxt for a built-in model (reference
examples/12_ConfigGenerator role).

    python tools/trtis_config_gen.py --model resnet50 --batch 8 \
        --max-batch 32 --instances 3 --preferred 4 8
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="resnet50")
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--max-batch", type=int, default=0)
    ap.add_argument("--instances", type=int, default=1)
    ap.add_argument("--preferred", type=int, nargs="*", default=())
    ap.add_argument("--queue-delay-us", type=int, default=100)
    args = ap.parse_args()

    from trtlab_amd.engine.planner import Planner
    from trtlab_amd.models import build_bert, build_resnet
    from trtlab_amd.rpc.trtis import model_config_pbtxt

    if args.model.startswith("resnet"):
        g = build_resnet(int(args.model[6:]), batch=args.batch, image=224,
                         seed=0)
    elif args.model == "bert":
        g = build_bert(batch=args.batch, seq=128, layers=12, seed=0)
    else:
        raise SystemExit(f"unknown model {args.model}")
    plan = Planner().compile(g)
    print(model_config_pbtxt(plan, args.model,
                             max_batch_size=args.max_batch,
                             instances=args.instances,
                             preferred_batch_sizes=args.preferred,
                             queue_delay_us=args.queue_delay_us))


if __name__ == "__main__":
    main()
