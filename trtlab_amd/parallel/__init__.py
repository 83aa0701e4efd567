"""trtlab_amd.parallel — owned RCCL collective layer over xGMI.

The reference has NO owned collective backend (MPI is
barriers-only for MPS benchmarks, examples/00_TensorRT/inference.cc:37-42).
This module is the MI355X-native addition: a first-class RCCL communicator
(csrc/runtime/comm.cpp — ncclCommInitRank/Broadcast/AllReduce on the
engine's streams, no torch.distributed in the data path), a file-based
unique-id rendezvous for single-node cliques (torchrun stays launcher-only),
weight broadcast at model load, and a failover-aware replica scheduler
behind the RPC load balancer.
"""
from __future__ import annotations

import os
import tempfile
import time
from typing import List, Optional, Sequence

# Reduction ops / dtypes — keep in sync with csrc/runtime/comm.h
OP_SUM, OP_PROD, OP_MAX, OP_MIN, OP_AVG = range(5)
DT_U8, DT_F16, DT_BF16, DT_F32, DT_F64, DT_I32 = range(6)

_UID_BYTES = 128  # NCCL_UNIQUE_ID_BYTES


def _rendezvous_path(world: int) -> str:
    """Rendezvous file path shared by all ranks of one launch. Keyed by the
    launcher's run identity (torchrun MASTER_PORT / TORCHELASTIC_RUN_ID)
    so concurrent jobs on one box don't collide."""
    key = os.environ.get("TORCHELASTIC_RUN_ID") or os.environ.get(
        "MASTER_PORT", "0")
    return os.path.join(tempfile.gettempdir(),
                        f"trtlab_rccl_uid_{key}_w{world}.bin")


def exchange_unique_id(rank: int, world: int, uid: Optional[bytes] = None,
                       path: Optional[str] = None,
                       timeout: float = 120.0) -> bytes:
    """File-based unique-id rendezvous: rank 0 publishes the 128-byte RCCL
    unique id atomically (write-tmp + rename); other ranks poll for it.

    This replaces torch.distributed's TCPStore for comm bring-up — the only
    out-of-band channel RCCL needs. Single-node scope (the driver's 8-GPU
    bench is one node; multi-node would pass `uid` via its own channel).
    """
    path = path or _rendezvous_path(world)
    t_entry = time.time()
    if rank == 0:
        if uid is None:
            raise ValueError("rank 0 must supply the uid to publish")
        try:  # drop any stale file from a previous launch first
            os.unlink(path)
        except FileNotFoundError:
            pass
        tmp = path + f".tmp.{os.getpid()}"
        with open(tmp, "wb") as f:
            f.write(uid)
            f.flush()
            os.fsync(f.fileno())
        os.replace(tmp, path)  # atomic publish
        return uid
    deadline = time.monotonic() + timeout
    while time.monotonic() < deadline:
        try:
            # reject stale files from earlier launches: all ranks start
            # within ms of each other (torchrun), so a uid published by THIS
            # launch has mtime ~t_entry; one from a previous run is older
            if os.path.getmtime(path) >= t_entry - 10.0:
                with open(path, "rb") as f:
                    data = f.read()
                if len(data) == _UID_BYTES:
                    return data
        except FileNotFoundError:
            pass
        time.sleep(0.05)
    raise TimeoutError(f"RCCL rendezvous timed out waiting for {path}")


def exchange_unique_id_tcp(rank: int, world: int,
                           uid: Optional[bytes] = None,
                           addr: Optional[str] = None,
                           port: Optional[int] = None,
                           timeout: float = 300.0) -> bytes:
    """TCP unique-id rendezvous for MULTI-NODE launches (the file path
    above needs a shared filesystem; this needs only the MASTER_ADDR
    reachability torchrun already assumes). rank 0 serves the 128-byte
    RCCL uid on (addr, port); every other rank connects with retry,
    sends its rank (4 bytes, for logging/validation), and reads the uid.
    Port defaults to MASTER_PORT+1 so torchrun's own TCPStore on
    MASTER_PORT is not disturbed."""
    import socket
    import struct

    addr = addr or os.environ.get("MASTER_ADDR", "127.0.0.1")
    port = port or int(os.environ.get("MASTER_PORT", "29500")) + 1
    deadline = time.monotonic() + timeout
    if rank == 0:
        if uid is None or len(uid) != _UID_BYTES:
            raise ValueError("rank 0 must supply the 128-byte uid")
        srv = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        srv.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        srv.bind(("0.0.0.0", port))
        srv.listen(world)
        srv.settimeout(1.0)
        served: set = set()
        try:
            while len(served) < world - 1:
                if time.monotonic() > deadline:
                    raise TimeoutError(
                        f"uid rendezvous: {len(served)}/{world - 1} peers "
                        f"after {timeout:.0f}s")
                try:
                    conn, _ = srv.accept()
                except socket.timeout:
                    continue
                with conn:
                    conn.settimeout(10.0)
                    peer = struct.unpack("!i", conn.recv(4))[0]
                    conn.sendall(uid)
                    served.add(peer)
        finally:
            srv.close()
        return uid
    last_err: Optional[Exception] = None
    while time.monotonic() < deadline:
        try:
            with socket.create_connection((addr, port), timeout=5.0) as c:
                c.sendall(struct.pack("!i", rank))
                data = b""
                while len(data) < _UID_BYTES:
                    chunk = c.recv(_UID_BYTES - len(data))
                    if not chunk:
                        raise ConnectionError("short uid read")
                    data += chunk
                return data
        except OSError as e:  # rank 0 not listening yet
            last_err = e
            time.sleep(0.2)
    raise TimeoutError(
        f"uid rendezvous: could not reach rank 0 at {addr}:{port} "
        f"({last_err})")


class Communicator:
    """Owned RCCL communicator (one per process; ranks may share a GPU —
    that is how the 2-rank proof runs within a 1-GPU lease).

    Wraps native trtlab::Communicator; collectives are stream-ordered
    against the engine's HIP streams when a stream is passed.
    """

    def __init__(self, rank: Optional[int] = None,
                 world: Optional[int] = None, device: Optional[int] = None,
                 uid: Optional[bytes] = None,
                 rendezvous_path: Optional[str] = None,
                 force: bool = False, rendezvous: str = "auto"):
        """force=True builds the native RCCL communicator even at world
        size 1 (a real self-clique: ncclCommInitRank + collectives execute
        on the GPU). Default world-1 behavior is no-op passthrough.

        rendezvous: "file" (single node, shared /tmp), "tcp" (multi-node:
        uid served from rank 0 on MASTER_PORT+1), or "auto" — tcp when
        MASTER_ADDR points off-host, else file. TRTLAB_RDZV overrides.

        NOTE: this RCCL build rejects two ranks on ONE device
        ("Duplicate GPU detected", ncclInvalidUsage) — a genuine N>1
        clique needs N distinct GPUs; the world-1 forced clique is the
        deepest single-GPU proof available."""
        from trtlab_amd import native

        C = native()
        self.rank = int(os.environ.get("RANK", "0")) if rank is None else rank
        self.world = (int(os.environ.get("WORLD_SIZE", "1"))
                      if world is None else world)
        if device is None:
            device = int(os.environ.get("LOCAL_RANK", str(self.rank)))
            ndev = C.hip.device_count()
            if ndev > 0:
                device = device % ndev  # multi-rank single-GPU proof mode
        self.device = device
        if self.world <= 1 and not force:
            self._comm = None
            return
        if uid is None:
            uid = C.comm.unique_id() if self.rank == 0 else None
            if self.world > 1:
                mode = os.environ.get("TRTLAB_RDZV", rendezvous)
                if mode == "auto":
                    master = os.environ.get("MASTER_ADDR", "127.0.0.1")
                    local = master in ("127.0.0.1", "localhost", "::1",
                                       os.uname().nodename)
                    mode = "file" if local else "tcp"
                if mode == "tcp":
                    uid = exchange_unique_id_tcp(self.rank, self.world, uid)
                else:
                    uid = exchange_unique_id(self.rank, self.world, uid,
                                             path=rendezvous_path)
        self._comm = C.comm.Communicator(self.rank, self.world, uid,
                                         self.device)

    # -- collectives (no-ops at world 1 so callers need no special-casing) --
    def broadcast(self, ptr: int, nbytes: int, root: int = 0,
                  stream: int = 0) -> None:
        if self._comm:
            self._comm.broadcast(ptr, nbytes, root, stream)

    def all_reduce(self, ptr: int, count: int, dtype: int = DT_F32,
                   op: int = OP_SUM, stream: int = 0) -> None:
        if self._comm:
            self._comm.all_reduce(ptr, count, dtype, op, stream)

    def all_gather(self, send_ptr: int, recv_ptr: int, bytes_per_rank: int,
                   stream: int = 0) -> None:
        if self._comm:
            self._comm.all_gather(send_ptr, recv_ptr, bytes_per_rank, stream)

    def reduce_scatter(self, send_ptr: int, recv_ptr: int,
                       count_per_rank: int, dtype: int = DT_F32,
                       op: int = OP_SUM, stream: int = 0) -> None:
        if self._comm:
            self._comm.reduce_scatter(send_ptr, recv_ptr, count_per_rank,
                                      dtype, op, stream)

    def send(self, ptr: int, nbytes: int, peer: int, stream: int = 0) -> None:
        if self._comm:
            self._comm.send(ptr, nbytes, peer, stream)

    def recv(self, ptr: int, nbytes: int, peer: int, stream: int = 0) -> None:
        if self._comm:
            self._comm.recv(ptr, nbytes, peer, stream)

    def barrier(self) -> None:
        if self._comm:
            self._comm.barrier()

    def all_reduce_scalar(self, v: float, op: int = OP_MAX) -> float:
        return self._comm.all_reduce_scalar(v, op) if self._comm else v

    def synchronize(self) -> None:
        if self._comm:
            self._comm.stream_synchronize()

    def close(self) -> None:
        self._comm = None  # native dtor runs ncclCommDestroy


def broadcast_weights(engine, comm: Communicator, src_rank: int = 0) -> None:
    """Broadcast the engine's device weight blob from src_rank over RCCL.

    One fused in-place broadcast of the whole blob directly on the engine's
    weight memory — no staging tensor, no torch (bigger transfers amortize
    per-link xGMI latency better than per-tensor messages).
    """
    if comm is None or comm._comm is None:
        return  # world-1 no-op comm (a forced self-clique still broadcasts)
    comm.broadcast(engine.engine.weights_ptr, engine.engine.weight_bytes,
                   root=src_rank)
    comm.synchronize()


# ---------------------------------------------------------------------------
# Replica scheduling (reference 00_TensorRT --replicas round-robin,
# inference.cc:227-230 — upgraded: queue-depth-aware pick + failover).
class ReplicaGroup:
    """Schedules requests over per-GPU engine replicas inside one process.

    Pick = least-outstanding among healthy replicas (falls back to round-
    robin on ties). A replica whose submit raises is marked unhealthy and
    skipped; mark_healthy() re-admits it (health RPC / operator action).
    """

    def __init__(self, plan=None, devices: Sequence[int] = (), engines=None):
        if engines is None:
            from trtlab_amd.engine.runtime import NativeEngine

            engines = [NativeEngine(plan, device=d) for d in devices]
        self.engines: List = list(engines)
        self.outstanding = [0] * len(self.engines)
        self.healthy = [True] * len(self.engines)
        self._next = 0

    def next_index(self) -> int:
        live = [i for i, h in enumerate(self.healthy) if h]
        if not live:
            raise RuntimeError("ReplicaGroup: no healthy replicas")
        best = min(live, key=lambda i: (self.outstanding[i],
                                        (i - self._next) % len(self.engines)))
        self._next = (best + 1) % len(self.engines)
        return best

    def next_engine(self):
        return self.engines[self.next_index()]

    def acquire(self) -> int:
        i = self.next_index()
        self.outstanding[i] += 1
        return i

    def release(self, i: int) -> None:
        self.outstanding[i] = max(0, self.outstanding[i] - 1)

    def mark_unhealthy(self, i: int) -> None:
        self.healthy[i] = False

    def mark_healthy(self, i: int) -> None:
        self.healthy[i] = True
