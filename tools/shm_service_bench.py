#!/usr/bin/env python3
"""Service-level benchmark over the POSIX-shm zero-copy transport
(InferRequest shm_name/shm_size): removes the 2.4 MB/request protobuf
serialization that bounds the plain path."""
import os
import subprocess
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))

import numpy as np

env = dict(os.environ)
server = subprocess.Popen(
    [sys.executable, "examples/inference_server.py", "--model", "resnet50",
     "--batch", "8", "--port", "50953", "--contexts", "3"],
    cwd=str(ROOT), env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
    text=True)
try:
    from trtlab_amd.rpc import HealthRequest, HealthResponse, SyncClient
    from trtlab_amd.rpc.remote import RemoteInferenceManager

    deadline = time.time() + 180
    while True:
        try:
            c = SyncClient("127.0.0.1:50953")
            r = c.call("trtlab.Health", "Check", HealthRequest(),
                       HealthResponse, timeout=2)
            if r.ready:
                break
        except Exception:
            pass
        if time.time() > deadline:
            raise RuntimeError("server did not come up")
        time.sleep(1)

    mgr = RemoteInferenceManager("127.0.0.1:50953")
    runner = mgr.infer_runner("resnet50", use_shm=True)  # pooled segments
    batch = np.random.randn(8, 224, 224, 3).astype(np.float16)

    # pipelined: keep `depth` requests in flight
    for depth in (1, 8, 32):
        for _ in range(20):
            runner.infer(batch).result(30)  # warm
        lat = []
        n = 300
        t0 = time.perf_counter()
        inflight = [runner.infer(batch) for _ in range(depth)]
        done = 0
        while done < n:
            f = inflight.pop(0)
            ts = time.perf_counter()
            f.result(30)
            done += 1
            if depth == 1:
                lat.append((time.perf_counter() - ts))
            if done + len(inflight) < n:
                inflight.append(runner.infer(batch))
        dt = time.perf_counter() - t0
        infs = n * 8 / dt
        extra = ""
        if depth == 1:
            extra = f"  (sync loop)"
        print(f"shm depth={depth:>2}: {infs:8.0f} inf/s{extra}", flush=True)
    runner.close()
    mgr.close()
finally:
    server.terminate()
    server.wait(10)
