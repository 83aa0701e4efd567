"""Multi-process distributed tests on CPU (gloo, world_size=2): exercise the
bench's rendezvous/broadcast/barrier flow without GPUs so the RCCL path is
correct by construction (the driver runs the real multi-GPU bench)."""
import os
import subprocess
import sys
import tempfile
from pathlib import Path

import numpy as np
import pytest

ROOT = Path(__file__).resolve().parent.parent

_WORKER = r"""
import os, sys
import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, os.environ["TRTLAB_ROOT"])

rank = int(os.environ["RANK"])
world = int(os.environ["WORLD_SIZE"])
dist.init_process_group("gloo")

# 1) weight-blob broadcast flow (CPU analogue of parallel.broadcast_weights)
from trtlab_amd.models import build_resnet
from trtlab_amd.engine.planner import Planner

g = build_resnet(50, batch=1, image=64, seed=0, calibrate=False)
plan = Planner().compile(g)
blob = torch.from_numpy(plan.weights.copy())
if rank != 0:
    blob.zero_()
dist.broadcast(blob, src=0)
assert np.array_equal(blob.numpy(), plan.weights), "broadcast mismatch"

# 2) timed-region choreography: barrier + max-of-elapsed all-reduce
import time
dist.barrier()
t = torch.tensor([0.1 * (rank + 1)], dtype=torch.float64)
dist.all_reduce(t, op=dist.ReduceOp.MAX)
assert abs(t.item() - 0.1 * world) < 1e-9
dist.barrier()
dist.destroy_process_group()
print(f"rank {rank} OK", flush=True)
"""


def test_gloo_world2_bench_flow(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(_WORKER)
    env = dict(os.environ)
    env["TRTLAB_ROOT"] = str(ROOT)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29611", str(script)],
        capture_output=True, text=True, timeout=240, env=env, cwd=str(ROOT))
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    assert r.stdout.count("OK") == 2
