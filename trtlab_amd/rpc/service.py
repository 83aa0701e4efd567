"""Inference gRPC service: wires an InferenceManager/InferRunner into the
RPC layer — the reference's 02_TensorRT_GRPC server (FlowersContext::
ExecuteRPC, server.cc:150-183) with its compute/request duration split and
prometheus metrics, rebuilt for the native MI355X engine.
"""
from __future__ import annotations

import asyncio
import time
from typing import Dict, Optional

import numpy as np

from trtlab_amd.core import Resources
from trtlab_amd.rpc.proto import (HealthRequest, HealthResponse, InferRequest,
                                  InferResponse)
from trtlab_amd.rpc.server import AsyncService
from trtlab_amd.utils import log


class InferenceResources(Resources):
    """Resources handle injected into RPC contexts (reference server.cc:287:
    FlowersResources{InferenceManager})."""

    def __init__(self, manager):
        self.manager = manager  # engine.runtime.InferenceManager
        self.runners: Dict[str, object] = {}

    def runner(self, model: str):
        if model not in self.runners:
            self.runners[model] = self.manager.infer_runner(model)
        return self.runners[model]


class InferenceService(AsyncService):
    """`trtlab.Inference/Compute` unary service. The RPC handler hops off
    the event loop into the InferRunner's pre/hip/post pipeline (reference:
    never block the CQ thread — server.cc:155 cuda_pool.enqueue) and
    reports compute vs request ms (server.cc:175-177)."""

    def __init__(self, resources: InferenceResources, metrics=None):
        super().__init__("trtlab.Inference", resources)
        self.metrics = metrics
        from collections import OrderedDict

        # name -> SharedMemory (pooled clients); true LRU: hits move to the
        # back, eviction pops the FRONT (oldest) so stale segments from dead
        # clients age out instead of thrashing the newest mapping
        self._shm_cache: "OrderedDict[str, object]" = OrderedDict()
        self.register_unary("Compute", self._compute, InferRequest,
                            InferResponse)

        async def health(request, context, _r):
            return HealthResponse(ready=True, status="serving")

        hs = AsyncService("trtlab.Health")
        hs.register_unary("Check", health, HealthRequest, HealthResponse)
        self.health_service = hs

    async def _compute(self, request: InferRequest, context, resources):
        t_start = time.monotonic()
        runner = resources.runner(request.model)
        plan = resources.manager.get_model(request.model).plan
        dtype = np.float16 if request.dtype in ("", "f16", "float16") else np.dtype(request.dtype)
        shape = tuple(request.shape) or plan.input_shape
        if request.inputs:
            # named multi-binding request (reference carves N addresses per
            # model; TRTIS names request inputs the same way)
            np_dt = {"f16": np.float16, "i32": np.int32, "f32": np.float32,
                     "bf16": np.int16, "i8": np.int8, "": np.float16}
            batch = {
                t.name: np.frombuffer(t.data, dtype=np_dt[t.dtype]).reshape(
                    tuple(t.shape)) for t in request.inputs}
        elif request.shm_name:
            # zero-copy local transport: the tensor lives in POSIX shared
            # memory (reference's SysV shm input path, 02 server.cc:159).
            # Mappings are cached by name: clients pool and reuse segments
            # (rpc.client.ShmPool), so re-mmapping per request is waste.
            from multiprocessing import shared_memory

            def read_shm():
                shm = self._shm_cache.get(request.shm_name)
                if shm is None:
                    shm = shared_memory.SharedMemory(name=request.shm_name)
                    if len(self._shm_cache) >= 64:  # bounded, evict oldest
                        self._shm_cache.popitem(last=False)[1].close()
                    self._shm_cache[request.shm_name] = shm
                else:
                    self._shm_cache.move_to_end(request.shm_name)
                return np.frombuffer(
                    shm.buf[:int(request.shm_size)], dtype=dtype
                ).reshape(shape).copy()

            try:
                batch = read_shm()
            except Exception:
                # stale cached mapping (client restarted): drop + reopen
                old = self._shm_cache.pop(request.shm_name, None)
                if old is not None:
                    old.close()
                batch = read_shm()
        else:
            batch = np.frombuffer(request.input, dtype=dtype).reshape(shape)

        t_compute = time.monotonic()
        fut = runner.infer(batch)
        out = await asyncio.wrap_future(fut)
        compute_ms = (time.monotonic() - t_compute) * 1e3
        request_ms = (time.monotonic() - t_start) * 1e3
        if self.metrics:
            self.metrics.observe(compute_ms, request_ms)
        if isinstance(out, dict):  # multi-output model -> named tensors
            from trtlab_amd.rpc.proto import NamedTensor

            prim = out[plan.outputs[0]["name"]]
            return InferResponse(
                output=prim.tobytes(), shape=list(prim.shape), dtype="f16",
                batch_id=request.batch_id, compute_ms=compute_ms,
                request_ms=request_ms,
                outputs=[NamedTensor(name=k, data=v.tobytes(),
                                     shape=list(v.shape),
                                     dtype=str(v.dtype).replace("float", "f"))
                         for k, v in out.items()])
        return InferResponse(
            output=out.tobytes(), shape=list(out.shape), dtype="f16",
            batch_id=request.batch_id, compute_ms=compute_ms,
            request_ms=request_ms)
