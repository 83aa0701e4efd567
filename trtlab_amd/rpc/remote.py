"""Remote inference manager: the client-side mirror of InferenceManager
(reference PyRemoteInferenceManager, pybind/trtlab/infer.cc:~83 — a gRPC
client exposing the same infer() future surface as the local runner)."""
from __future__ import annotations

from concurrent.futures import Future
from typing import Optional, Tuple

import numpy as np

from trtlab_amd.rpc.client import AsyncClient, ShmInput, ShmPool
from trtlab_amd.rpc.proto import (HealthRequest, HealthResponse, InferRequest,
                                  InferResponse)


class RemoteInferRunner:
    """infer(batch) -> Future[np.ndarray], like the local InferRunner."""

    def __init__(self, manager: "RemoteInferenceManager", model: str,
                 use_shm: bool = False, shm_depth: int = 8):
        self._m = manager
        self.model = model
        self.use_shm = use_shm  # zero-copy local transport (pooled)
        self._shm_depth = shm_depth
        self._pool: ShmPool | None = None

    def infer(self, batch: np.ndarray) -> Future:
        batch = np.ascontiguousarray(batch, np.float16)
        seg = name = None
        size = 0
        if self.use_shm:
            if self._pool is None:
                self._pool = ShmPool(batch.nbytes, depth=self._shm_depth)
            seg, name, size = self._pool.checkout(batch)
        req = InferRequest(
            model=self.model, shape=list(batch.shape), dtype="f16",
            input=b"" if seg is not None else batch.tobytes(),
            shm_name=name or "", shm_size=size)
        inner = self._m._client.call("trtlab.Inference", "Compute", req,
                                     InferResponse, timeout=self._m.timeout)
        out: Future = Future()

        def done(f):
            if seg is not None:
                self._pool.release(seg)
            exc = f.exception()
            if exc is not None:
                out.set_exception(exc)
                return
            r = f.result()
            arr = np.frombuffer(r.output, dtype=np.float16).reshape(
                tuple(r.shape))
            out.set_result(arr)

        inner.add_done_callback(done)
        return out

    def close(self):
        if self._pool is not None:
            self._pool.close()
            self._pool = None


class RemoteInferenceManager:
    """Connects to a trtlab.Inference server; hands out runners."""

    def __init__(self, target: str, timeout: float = 60.0):
        self.target = target
        self.timeout = timeout
        self._client = AsyncClient(target)

    def ready(self, timeout: float = 5.0) -> bool:
        try:
            r = self._client.call("trtlab.Health", "Check", HealthRequest(),
                                  HealthResponse, timeout=timeout).result(
                                      timeout + 1)
            return bool(r.ready)
        except Exception:
            return False

    def infer_runner(self, model: str, use_shm: bool = False) -> RemoteInferRunner:
        return RemoteInferRunner(self, model, use_shm=use_shm)

    def close(self):
        self._client.close()
