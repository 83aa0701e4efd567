"""Multi-process distributed tests on CPU (gloo, world_size=2): exercise the
bench's rendezvous/broadcast/barrier flow without GPUs so the RCCL path is
correct by construction (the driver runs the real multi-GPU bench)."""
import os
import subprocess
import sys
import tempfile
import time
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


# ---------------------------------------------------------------------------
# Owned-comm support logic that runs without a GPU: the unique-id rendezvous
# (trtlab_amd.parallel.exchange_unique_id) and the replica scheduler.

def test_rendezvous_publish_and_read(tmp_path):
    import threading

    from trtlab_amd.parallel import exchange_unique_id

    path = str(tmp_path / "uid.bin")
    uid = bytes(range(128))
    got = {}

    def reader():
        got["uid"] = exchange_unique_id(1, 2, path=path, timeout=10)

    t = threading.Thread(target=reader)
    t.start()
    exchange_unique_id(0, 2, uid=uid, path=path)
    t.join(timeout=10)
    assert got["uid"] == uid


def test_rendezvous_rejects_stale_file(tmp_path):
    import os
    import time

    from trtlab_amd.parallel import exchange_unique_id

    path = str(tmp_path / "uid.bin")
    with open(path, "wb") as f:
        f.write(b"\x01" * 128)
    old = time.time() - 3600
    os.utime(path, (old, old))  # a file from a previous launch
    with pytest.raises(TimeoutError):
        exchange_unique_id(1, 2, path=path, timeout=0.5)


def test_rendezvous_rank0_requires_uid(tmp_path):
    from trtlab_amd.parallel import exchange_unique_id

    with pytest.raises(ValueError):
        exchange_unique_id(0, 2, path=str(tmp_path / "x.bin"))


def test_communicator_world1_is_noop():
    from trtlab_amd.parallel import Communicator

    c = Communicator(rank=0, world=1, device=0)
    c.broadcast(0, 0)
    c.barrier()
    assert c.all_reduce_scalar(3.5) == 3.5
    c.close()


def test_replica_group_least_outstanding_and_failover():
    from trtlab_amd.parallel import ReplicaGroup

    g = ReplicaGroup(engines=["e0", "e1", "e2"])
    a = g.acquire()
    b = g.acquire()
    c = g.acquire()
    assert sorted([a, b, c]) == [0, 1, 2]  # spreads across all replicas
    g.release(b)
    assert g.acquire() == b  # least-outstanding wins
    # failover: unhealthy replicas are never picked
    g.mark_unhealthy(0)
    picks = {g.acquire() for _ in range(4)}
    assert 0 not in picks
    g.mark_healthy(0)
    g.outstanding = [0, 5, 5]
    assert g.next_index() == 0
    # all down -> loud failure
    g2 = ReplicaGroup(engines=["x"])
    g2.mark_unhealthy(0)
    with pytest.raises(RuntimeError):
        g2.next_index()


def _tcp_rdzv_worker(rank, world, port, q):
    import os

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    from trtlab_amd.parallel import exchange_unique_id_tcp

    uid = bytes(range(128)) if rank == 0 else None
    got = exchange_unique_id_tcp(rank, world, uid, addr="127.0.0.1",
                                 port=port, timeout=30.0)
    q.put((rank, got))


def test_tcp_uid_rendezvous_three_ranks():
    """Multi-node bring-up channel: rank 0 serves the 128-byte RCCL uid
    over TCP (MASTER_PORT+1); peers connect with retry. Pure transport
    test (no GPU): three processes, all must read rank 0's exact bytes —
    including a peer that starts BEFORE the server is listening."""
    import multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29871
    world = 3
    # start a non-zero rank FIRST to exercise the connect-retry path
    p1 = ctx.Process(target=_tcp_rdzv_worker, args=(1, world, port, q))
    p1.start()
    time.sleep(0.3)
    p0 = ctx.Process(target=_tcp_rdzv_worker, args=(0, world, port, q))
    p2 = ctx.Process(target=_tcp_rdzv_worker, args=(2, world, port, q))
    p0.start()
    p2.start()
    got = {}
    for _ in range(world):
        r, uid = q.get(timeout=60)
        got[r] = uid
    for p in (p0, p1, p2):
        p.join(timeout=30)
        assert p.exitcode == 0
    expect = bytes(range(128))
    assert got == {0: expect, 1: expect, 2: expect}


def test_tcp_uid_rendezvous_timeout():
    """A peer with no rank 0 to reach fails with a clear TimeoutError."""
    from trtlab_amd.parallel import exchange_unique_id_tcp

    with pytest.raises(TimeoutError, match="rank 0"):
        exchange_unique_id_tcp(1, 2, addr="127.0.0.1", port=29899,
                               timeout=1.0)
