#!/usr/bin/env python3
"""60 s constant-rate soak over the pooled-shm transport."""
import os
import subprocess
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))

import numpy as np

RATE = float(sys.argv[1]) if len(sys.argv) > 1 else 1200.0  # req/s (x8 inf)
DUR = float(sys.argv[2]) if len(sys.argv) > 2 else 60.0

env = dict(os.environ)
server = subprocess.Popen(
    [sys.executable, "examples/inference_server.py", "--model", "resnet50",
     "--batch", "8", "--port", "50955", "--contexts", "3"],
    cwd=str(ROOT), env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
    text=True)
try:
    from trtlab_amd.rpc import HealthRequest, HealthResponse, SyncClient
    from trtlab_amd.rpc.remote import RemoteInferenceManager

    deadline = time.time() + 180
    while True:
        try:
            r = SyncClient("127.0.0.1:50955").call(
                "trtlab.Health", "Check", HealthRequest(), HealthResponse,
                timeout=2)
            if r.ready:
                break
        except Exception:
            pass
        if time.time() > deadline:
            raise RuntimeError("server did not come up")
        time.sleep(1)

    mgr = RemoteInferenceManager("127.0.0.1:50955")
    runner = mgr.infer_runner("resnet50", use_shm=True, )
    runner._shm_depth = 64
    batch = np.random.randn(8, 224, 224, 3).astype(np.float16)
    for _ in range(20):
        runner.infer(batch).result(30)

    lat = []
    errors = 0
    sent = 0
    inflight = []
    t0 = time.perf_counter()
    period = 1.0 / RATE
    while (now := time.perf_counter()) - t0 < DUR:
        target = t0 + sent * period
        if now < target:
            time.sleep(min(target - now, 0.001))
            continue
        ts = time.perf_counter()
        f = runner.infer(batch)
        f._t0 = ts
        inflight.append(f)
        sent += 1
        done = [f for f in inflight if f.done()]
        for f in done:
            inflight.remove(f)
            try:
                f.result(0)
                lat.append(time.perf_counter() - f._t0)
            except Exception:
                errors += 1
    for f in inflight:
        try:
            f.result(30)
            lat.append(time.perf_counter() - f._t0)
        except Exception:
            errors += 1
    dt = time.perf_counter() - t0
    lat_ms = np.array(lat) * 1e3
    print(f"soak {RATE:.0f} req/s x {DUR:.0f}s (pooled shm): "
          f"{len(lat)}/{sent} ok, {errors} errors, "
          f"{len(lat)*8/dt:.0f} inf/s, p50 {np.percentile(lat_ms,50):.1f} "
          f"p99 {np.percentile(lat_ms,99):.1f} ms", flush=True)
    runner.close()
    mgr.close()
finally:
    server.terminate()
    server.wait(10)
