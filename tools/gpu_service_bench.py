"""End-to-end gRPC service measurement on one GPU box: start the inference
server, drive it with the sync client and the siege load generator, print
service-level inf/sec + latency (reference config: 02_TensorRT_GRPC
client-sync.x measured 371.7 inf/s on V100)."""
import os
import subprocess
import sys
import time
from pathlib import Path

ROOT = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(ROOT))

env = dict(os.environ)
server = subprocess.Popen(
    [sys.executable, "examples/inference_server.py", "--model", "resnet50",
     "--batch", "8", "--port", "50951", "--metrics-port", "50978",
     "--contexts", "3"],
    cwd=str(ROOT), env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
    text=True)
try:
    # wait for readiness via health check
    from trtlab_amd.rpc import HealthRequest, HealthResponse, SyncClient

    deadline = time.time() + 180
    while True:
        try:
            c = SyncClient("127.0.0.1:50951")
            r = c.call("trtlab.Health", "Check", HealthRequest(),
                       HealthResponse, timeout=2)
            c.close()
            if r.ready:
                break
        except Exception:
            if time.time() > deadline:
                raise RuntimeError("server did not come up")
            time.sleep(2)
    print("server ready", flush=True)

    r = subprocess.run(
        [sys.executable, "examples/client.py", "--target", "127.0.0.1:50951",
         "--count", "200", "--mode", "sync"],
        cwd=str(ROOT), capture_output=True, text=True, timeout=300)
    print("== sync client ==\n" + r.stdout + r.stderr, flush=True)

    r = subprocess.run(
        [sys.executable, "examples/client.py", "--target", "127.0.0.1:50951",
         "--count", "300", "--mode", "async"],
        cwd=str(ROOT), capture_output=True, text=True, timeout=300)
    print("== async client ==\n" + r.stdout + r.stderr, flush=True)

    r = subprocess.run(
        [sys.executable, "examples/siege.py", "--target", "127.0.0.1:50951",
         "--rate", "300", "--seconds", "5", "--max-outstanding", "64"],
        cwd=str(ROOT), capture_output=True, text=True, timeout=300)
    print("== siege @300 req/s ==\n" + r.stdout + r.stderr, flush=True)

    # metrics endpoint spot check
    import urllib.request

    txt = urllib.request.urlopen("http://127.0.0.1:50978/metrics",
                                 timeout=5).read().decode()
    keep = [l for l in txt.splitlines()
            if l.startswith("trtlab_") and ("count" in l or "sum" in l)]
    print("== prometheus ==\n" + "\n".join(keep[:12]), flush=True)
finally:
    server.terminate()
    try:
        server.wait(timeout=10)
    except subprocess.TimeoutExpired:
        server.kill()
