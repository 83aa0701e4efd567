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
    ap.add_argument("--workers", type=int, default=1,
                    help="N server PROCESSES sharing the port via "
                         "SO_REUSEPORT — each with its own interpreter (no "
                         "shared GIL) and its own engine contexts; the "
                         "remote-protobuf scale-out past Python "
                         "serialization (reference nvrpc runs N CQ threads, "
                         "executor.h:39)")
    args = ap.parse_args()

    # fork BEFORE any HIP/engine initialization (forking a process with a
    # live HIP context is undefined); each worker then builds its own
    # engine + contexts and binds the same port
    worker_id = 0
    if args.workers > 1:
        import os

        for i in range(1, args.workers):
            if os.fork() == 0:
                worker_id = i
                # die with the parent (no orphan workers holding the GPU)
                import ctypes
                import signal

                libc = ctypes.CDLL("libc.so.6", use_errno=True)
                libc.prctl(1, signal.SIGTERM)  # PR_SET_PDEATHSIG
                break
        if args.metrics_port and worker_id:
            args.metrics_port += worker_id  # one exposer per worker

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
    from trtlab_amd.rpc.trtis import TrtisService

    server.register_service(TrtisService(resources).service)
    # GC tuning for the serving loop: the engine/plan object graph is
    # permanent — freeze it out of collection, and raise gen0 threshold so
    # request-object churn (1.2 MB protobufs) doesn't trigger frequent
    # collections whose pauses land in the latency tail
    import gc

    gc.collect()
    gc.freeze()
    gc.set_threshold(50000, 100, 100)

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
