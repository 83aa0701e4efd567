"""RPC server framework (reference: trtlab/nvrpc server.h / service.h /
life_cycle_*.h / executor.h, rebuilt on grpc.aio).

Lifecycles map 1:1 to the reference's:
  UnaryService      ~ LifeCycleUnary      (life_cycle_unary.h:33)
  StreamingService  ~ LifeCycleStreaming  (life_cycle_streaming.h:61)
  BatchingService   ~ LifeCycleBatching   (life_cycle_batching.h:75) — a
                      unary front that dynamic-batches into a Dispatcher
The Executor role (N threads <-> N CQs, executor.h:39) is played by the
asyncio event loop running in a dedicated thread so synchronous apps get
the same Run()/AsyncStart()/Shutdown() surface as the reference Server
(server.h:40).
"""
from __future__ import annotations

import asyncio
import threading
import time
from concurrent.futures import Future
from typing import Any, Callable, Dict, List, Optional

import grpc
import grpc.aio

from trtlab_amd.core import Dispatcher, Resources
from trtlab_amd.utils import log


class _RpcDef:
    def __init__(self, name: str, handler, kind: str, req_cls, resp_cls):
        self.name = name
        self.handler = handler
        self.kind = kind  # 'unary' | 'stream_stream'
        self.req_cls = req_cls
        self.resp_cls = resp_cls


class AsyncService:
    """A named gRPC service assembled from registered RPCs (reference
    service.h:35 RegisterRPC)."""

    def __init__(self, full_name: str, resources: Optional[Resources] = None):
        self.full_name = full_name  # e.g. "trtlab.Inference"
        self.resources = resources
        self._rpcs: List[_RpcDef] = []

    def register_unary(self, method: str, handler, req_cls, resp_cls):
        """handler: async def (request, context, resources) -> response"""
        self._rpcs.append(_RpcDef(method, handler, "unary", req_cls, resp_cls))

    def register_streaming(self, method: str, handler, req_cls, resp_cls):
        """handler: async generator (request_iter, context, resources)"""
        self._rpcs.append(_RpcDef(method, handler, "stream_stream", req_cls,
                                  resp_cls))

    def _generic_handler(self) -> grpc.GenericRpcHandler:
        handlers: Dict[str, grpc.RpcMethodHandler] = {}
        for r in self._rpcs:
            if r.kind == "unary":
                async def u(request, context, _r=r):
                    return await _r.handler(request, context, self.resources)

                handlers[r.name] = grpc.unary_unary_rpc_method_handler(
                    u, request_deserializer=r.req_cls.FromString,
                    response_serializer=lambda m: m.SerializeToString())
            else:
                async def s(request_iter, context, _r=r):
                    async for resp in _r.handler(request_iter, context,
                                                 self.resources):
                        yield resp

                handlers[r.name] = grpc.stream_stream_rpc_method_handler(
                    s, request_deserializer=r.req_cls.FromString,
                    response_serializer=lambda m: m.SerializeToString())
        return grpc.method_handlers_generic_handler(self.full_name, handlers)


class UnaryService(AsyncService):
    pass


class StreamingService(AsyncService):
    pass


class BatchingService(AsyncService):
    """Unary front that collects requests into batches via the core
    Dispatcher (reference life_cycle_batching.h + 03_Batching example):
    ExecuteRPC sees vector<Request> -> vector<Response>."""

    def __init__(self, full_name: str, method: str, req_cls, resp_cls,
                 compute_batch_fn: Callable[[List[Any]], List[Any]],
                 max_batch_size: int = 8, timeout_s: float = 0.005,
                 resources: Optional[Resources] = None, workers: int = 2):
        super().__init__(full_name, resources)
        self._dispatcher = Dispatcher(max_batch_size, timeout_s,
                                      compute_batch_fn, workers=workers)

        async def handler(request, context, _resources):
            fut = self._dispatcher.enqueue(request)
            return await asyncio.wrap_future(fut)

        self.register_unary(method, handler, req_cls, resp_cls)

    def shutdown(self):
        self._dispatcher.shutdown()


class Server:
    """Owns the grpc.aio server + its event-loop thread (reference
    server.h:40: RegisterAsyncService, Run(timeout, control_fn),
    AsyncStart, Shutdown)."""

    def __init__(self, address: str = "0.0.0.0:50051"):
        self.address = address
        self._services: List[AsyncService] = []
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._thread: Optional[threading.Thread] = None
        self._server: Optional[grpc.aio.Server] = None
        self._started = threading.Event()
        self._stopped = threading.Event()
        self.port: Optional[int] = None

    def register_service(self, svc: AsyncService) -> AsyncService:
        self._services.append(svc)
        return svc

    # ------------------------------------------------------------- control
    def async_start(self) -> None:
        if self._thread:
            return
        self._thread = threading.Thread(target=self._run_loop, daemon=True,
                                        name="rpc-server")
        self._thread.start()
        if not self._started.wait(timeout=10):
            raise RuntimeError("gRPC server failed to start")

    def _run_loop(self):
        self._loop = asyncio.new_event_loop()
        asyncio.set_event_loop(self._loop)
        try:
            self._loop.run_until_complete(self._serve())
        finally:
            self._loop.close()

    async def _serve(self):
        # SO_REUSEPORT: N worker PROCESSES can bind the same port and the
        # kernel load-balances connections — the scale-out past the GIL
        # that the reference gets from N completion-queue threads
        # (nvrpc/executor.h:39). See examples/inference_server.py --workers.
        self._server = grpc.aio.server(options=[("grpc.so_reuseport", 1)])
        for svc in self._services:
            self._server.add_generic_rpc_handlers((svc._generic_handler(),))
        self.port = self._server.add_insecure_port(self.address)
        await self._server.start()
        log.info("rpc server listening on %s (port %d)", self.address, self.port)
        self._started.set()
        await self._server.wait_for_termination()
        self._stopped.set()

    def run(self, control_interval_s: float = 2.0,
            control_fn: Optional[Callable[[], None]] = None):
        """Blocking run with a periodic control lambda (reference
        Server::Run's 2 s NVML-power loop, 02 server.cc:322-330)."""
        self.async_start()
        try:
            while not self._stopped.is_set():
                time.sleep(control_interval_s)
                if control_fn:
                    control_fn()
        except KeyboardInterrupt:
            pass
        finally:
            self.shutdown()

    def shutdown(self, grace: float = 1.0):
        if self._loop and self._server:
            fut = asyncio.run_coroutine_threadsafe(
                self._server.stop(grace), self._loop)
            try:
                fut.result(timeout=grace + 5)
            except Exception:
                pass
        if self._thread:
            self._thread.join(timeout=5)
            self._thread = None
