#!/usr/bin/env python3
"""Request-forwarding middleman (reference 04_Middleman /
Deployment/batcher.cc role): a unary front that relays inference
requests to a backend server and returns its responses, adding one
hop of (measured) latency. Useful as the insertion point for
cross-host batching, admission control, or A/B routing.

    python examples/middleman.py --backend 127.0.0.1:50052 --port 50070
"""
import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from trtlab_amd.rpc.middleman import ForwardingService  # noqa: E402
from trtlab_amd.rpc.server import Server  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--backend", default="127.0.0.1:50052")
    ap.add_argument("--port", type=int, default=50070)
    ap.add_argument("--service", default="trtlab.Inference")
    ap.add_argument("--method", default="Infer")
    args = ap.parse_args()

    fwd = ForwardingService(args.backend, args.service, args.method)
    srv = Server(f"0.0.0.0:{args.port}")
    srv.register_service(fwd.service)
    srv.async_start()
    print(f"middleman :{srv.port} -> {args.backend} "
          f"({args.service}/{args.method}); ctrl-c to stop", flush=True)
    try:
        while True:
            time.sleep(2)
            s = fwd.stats()
            print(f"  forwarded {s['requests']} "
                  f"(p50 hop {s['p50_ms']:.2f} ms)", flush=True)
    except KeyboardInterrupt:
        srv.shutdown()


if __name__ == "__main__":
    main()
