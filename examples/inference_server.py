#!/usr/bin/env python3
"""gRPC inference server (reference: examples/02_TensorRT_GRPC/src/server.cc).

Serves ResNet-50/152 or BERT-base over `trtlab.Inference/Compute` with
prometheus metrics (compute/request summaries, load ratio, GPU power) and a
health service. Replicates the engine across --devices GPUs behind an
in-process round-robin (reference's multi-process + envoy pattern collapsed
into one process).

  python examples/inference_server.py --model resnet50 --port 50051 \
      --devices 0 --contexts 4 --metrics-port 50078
"""
import argparse
import sys
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="resnet50",
                    choices=["resnet50", "resnet101", "resnet152", "bert"])
    ap.add_argument("--batch", type=int, default=8)
    ap.add_argument("--port", type=int, default=50051)
    ap.add_argument("--metrics-port", type=int, default=50078)
    ap.add_argument("--contexts", type=int, default=4)
    ap.add_argument("--devices", default="0",
                    help="comma-separated GPU ids for replica-per-GPU")
    ap.add_argument("--shared-arena", action="store_true",
                    help="serve all models from one growing best-fit device "
                         "pool (native DeviceArena); exports pool gauges")
    args = ap.parse_args()

    from trtlab_amd.engine.planner import Planner
    from trtlab_amd.engine.runtime import InferenceManager
    from trtlab_amd.models import build_bert, build_resnet
    from trtlab_amd.rpc.server import Server
    from trtlab_amd.rpc.service import InferenceResources, InferenceService
    from trtlab_amd.utils.metrics import Metrics

    if args.model == "bert":
        g = build_bert(batch=args.batch, seq=128, layers=12, seed=0)
    else:
        g = build_resnet(int(args.model.replace("resnet", "")),
                         batch=args.batch, seed=0)
    plan = Planner().compile(g)

    devices = [int(d) for d in args.devices.split(",")]
    managers = []
    for dev in devices:
        mgr = InferenceManager(max_contexts=args.contexts, device=dev,
                               shared_arena=args.shared_arena)
        mgr.register_model(args.model, plan)
        mgr.allocate_resources()
        managers.append(mgr)

    metrics = Metrics.initialize(args.metrics_port)
    resources = InferenceResources(managers[0]) if len(managers) == 1 else \
        _ReplicaResources(managers)
    svc = InferenceService(resources, metrics=metrics)

    server = Server(f"0.0.0.0:{args.port}")
    server.register_service(svc)
    server.register_service(svc.health_service)
    print(f"serving {args.model} b{args.batch} on :{args.port} "
          f"(devices {devices}, {args.contexts} contexts each; "
          f"metrics :{args.metrics_port})")
    def control():
        metrics.update_power(devices[0])
        metrics.update_arena(managers[0])

    server.run(control_interval_s=2.0, control_fn=control)


class _ReplicaResources:
    """Round-robin over per-GPU InferenceManagers."""

    def __init__(self, managers):
        from trtlab_amd.rpc.service import InferenceResources

        self.pools = [InferenceResources(m) for m in managers]
        self.manager = managers[0]
        self._i = 0

    def runner(self, model):
        self._i += 1
        return self.pools[self._i % len(self.pools)].runner(model)


if __name__ == "__main__":
    main()
