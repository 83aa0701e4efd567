"""trtlab_amd.parallel — RCCL over xGMI multi-GPU support.

The reference has NO owned collective backend (MPI is
barriers-only for MPS benchmarks). This module is the MI355X-native
addition: one process per GPU over torch.distributed (backend "nccl" IS
RCCL on ROCm), weight broadcast at model load, data-parallel replica
groups behind the RPC load balancer.
"""
from __future__ import annotations

import os
from typing import Optional


def init_distributed(backend: str = "nccl") -> tuple[int, int, int]:
    """Initialize torch.distributed from torchrun env; returns
    (rank, world_size, local_rank)."""
    import torch
    import torch.distributed as dist

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))
    if world > 1 and not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        if backend == "nccl":
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend)
    return rank, world, local_rank


def broadcast_weights(engine, src_rank: int = 0, device: int = 0) -> None:
    """Broadcast the engine's device weight blob from src_rank over RCCL.

    One fused broadcast of the whole blob (bigger transfers amortize the
    per-link xGMI latency better than per-tensor messages — SURVEY.md §2.9).
    The blob is copied device-to-device between torch's staging tensor and
    the engine's native buffer (same HIP address space).
    """
    import torch
    import torch.distributed as dist

    if not dist.is_initialized() or dist.get_world_size() == 1:
        return
    from trtlab_amd import native

    C = native()
    nbytes = engine.engine.weight_bytes
    staging = torch.empty(nbytes, dtype=torch.uint8, device=f"cuda:{device}")
    # local blob -> staging (D2D), broadcast, staging -> blob (D2D)
    C.memory.memcpy_d2d(staging.data_ptr(), engine.engine.weights_ptr, nbytes)
    torch.cuda.synchronize()
    dist.broadcast(staging, src=src_rank)
    torch.cuda.synchronize()
    C.memory.memcpy_d2d(engine.engine.weights_ptr, staging.data_ptr(), nbytes)
    C.hip.device_synchronize()


class ReplicaGroup:
    """Round-robin scheduler over per-GPU engine replicas inside one process
    (reference 00_TensorRT --replicas, inference.cc:227-230 — here each
    replica is a different GPU)."""

    def __init__(self, plan, devices):
        from trtlab_amd.engine.runtime import NativeEngine

        self.engines = [NativeEngine(plan, device=d) for d in devices]
        self._next = 0

    def next_engine(self):
        e = self.engines[self._next % len(self.engines)]
        self._next += 1
        return e
