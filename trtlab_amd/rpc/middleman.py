"""Forwarding middleman: a unary service that relays raw request bytes
to a backend gRPC server and returns the backend's raw response bytes.

Role of the reference's 04_Middleman / Deployment batcher front
(request forwarding between client and inference server). Operating on
raw bytes keeps the relay schema-agnostic — any unary method can be
proxied without importing its message classes — and adds a measured
hop-latency histogram for admission/monitoring decisions.
"""
from __future__ import annotations

import threading
import time
from typing import List

import grpc


class _RawBytes:
    """Message stand-in: serialization is the identity on bytes."""

    def __init__(self, data: bytes = b""):
        self.data = data

    def SerializeToString(self) -> bytes:  # noqa: N802 (grpc contract)
        return self.data

    @staticmethod
    def FromString(data: bytes) -> "_RawBytes":  # noqa: N802
        return _RawBytes(data)


class ForwardingService:
    """Unary front `<service>/<method>` relaying to the same method on
    the backend address."""

    def __init__(self, backend: str, service: str = "trtlab.Inference",
                 method: str = "Infer", timeout_s: float = 30.0):
        from trtlab_amd.rpc.server import AsyncService

        self.backend = backend
        self.timeout_s = timeout_s
        self._chan = grpc.insecure_channel(backend)
        self._call = self._chan.unary_unary(
            f"/{service}/{method}",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=_RawBytes.FromString)
        self._lat_ms: List[float] = []
        self._n = 0
        self._mu = threading.Lock()

        svc = AsyncService(service)

        async def handler(request, context, resources):
            import asyncio

            t0 = time.perf_counter()
            loop = asyncio.get_running_loop()
            resp = await loop.run_in_executor(
                None, lambda: self._call(request,
                                         timeout=self.timeout_s))
            dt = (time.perf_counter() - t0) * 1e3
            with self._mu:
                self._n += 1
                self._lat_ms.append(dt)
                if len(self._lat_ms) > 4096:
                    del self._lat_ms[:2048]
            return resp

        svc.register_unary(method, handler, _RawBytes, _RawBytes)
        self.service = svc

    def stats(self) -> dict:
        import numpy as np

        with self._mu:
            lat = list(self._lat_ms)
            n = self._n
        return dict(requests=n,
                    p50_ms=float(np.percentile(lat, 50)) if lat else 0.0,
                    p99_ms=float(np.percentile(lat, 99)) if lat else 0.0)

    def close(self):
        self._chan.close()
