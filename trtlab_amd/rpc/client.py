"""RPC clients: sync + async unary/streaming, and the siege constant-rate
load generator (reference client/client_unary.h, client_streaming*.h,
02_TensorRT_GRPC/src/siege.cc:226-258)."""
from __future__ import annotations

import asyncio
import queue
import threading
import time
from concurrent.futures import Future
from typing import Any, AsyncIterator, Callable, List, Optional

import grpc
import grpc.aio


class ShmInput:
    """Zero-copy local input: place the batch in POSIX shared memory and
    reference it by name in InferRequest.shm_name (reference
    SharedMemoryService / sysv_allocator.cc). Use as a context manager."""

    def __init__(self, batch):
        from multiprocessing import shared_memory
        import numpy as np

        arr = np.ascontiguousarray(batch)
        self.shm = shared_memory.SharedMemory(create=True, size=arr.nbytes)
        self.shm.buf[:arr.nbytes] = arr.tobytes()
        self.name = self.shm.name
        self.size = arr.nbytes

    def __enter__(self):
        return self

    def __exit__(self, *exc):
        self.close()
        return False

    def close(self):
        self.shm.close()
        try:
            self.shm.unlink()
        except FileNotFoundError:
            pass




class ShmPool:
    """Reusable POSIX shared-memory segments for the zero-copy transport:
    per-request segment creation (shm_open + mmap + unlink) measured as
    the dominant client-side cost of ShmInput (profiles/README), so a
    runner checks segments out of this pool and returns them when the
    response lands. Sized for `depth` concurrent requests."""

    def __init__(self, nbytes: int, depth: int = 8):
        import queue
        from multiprocessing import shared_memory

        self.nbytes = nbytes
        self._q = queue.Queue()
        self._all = []
        for _ in range(depth):
            shm = shared_memory.SharedMemory(create=True, size=nbytes)
            self._all.append(shm)
            self._q.put(shm)

    def checkout(self, arr) -> "tuple":
        import numpy as np

        a = np.ascontiguousarray(arr)
        assert a.nbytes <= self.nbytes
        shm = self._q.get()  # blocks when `depth` requests are in flight
        shm.buf[:a.nbytes] = a.tobytes()
        return shm, shm.name, a.nbytes

    def release(self, shm) -> None:
        self._q.put(shm)

    def close(self):
        while not self._q.empty():
            self._q.get_nowait()
        for shm in self._all:
            shm.close()
            try:
                shm.unlink()
            except FileNotFoundError:
                pass
        self._all.clear()


class SyncClient:
    """Blocking unary client over a shared channel."""

    def __init__(self, target: str):
        self.channel = grpc.insecure_channel(target)

    def call(self, service: str, method: str, request, resp_cls,
             timeout: Optional[float] = None):
        fn = self.channel.unary_unary(
            f"/{service}/{method}",
            request_serializer=lambda m: m.SerializeToString(),
            response_deserializer=resp_cls.FromString)
        return fn(request, timeout=timeout)

    def close(self):
        self.channel.close()


class AsyncClient:
    """Async unary/streaming client with its own event-loop thread, so
    synchronous code gets future-based Infer (reference client::Executor
    (client/executor.h:39) + ClientUnary's async_compute futures)."""

    def __init__(self, target: str):
        self.target = target
        self._loop = asyncio.new_event_loop()
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="rpc-client")
        self._ready = threading.Event()
        self._thread.start()
        self._ready.wait(timeout=10)

    def _run(self):
        asyncio.set_event_loop(self._loop)
        self.channel = grpc.aio.insecure_channel(self.target)
        self._ready.set()
        self._loop.run_forever()

    def call(self, service: str, method: str, request, resp_cls,
             timeout: Optional[float] = None) -> Future:
        async def do():
            fn = self.channel.unary_unary(
                f"/{service}/{method}",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=resp_cls.FromString)
            return await fn(request, timeout=timeout)

        return asyncio.run_coroutine_threadsafe(do(), self._loop)

    def stream(self, service: str, method: str, requests: List[Any],
               resp_cls) -> Future:
        """Bidirectional stream: send all requests, collect all responses."""
        async def do():
            fn = self.channel.stream_stream(
                f"/{service}/{method}",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=resp_cls.FromString)

            async def gen():
                for r in requests:
                    yield r

            return [resp async for resp in fn(gen())]

        return asyncio.run_coroutine_threadsafe(do(), self._loop)

    def open_stream(self, service: str, method: str, resp_cls,
                    on_response: Optional[Callable[[Any], None]] = None
                    ) -> "StreamingCall":
        """Incremental bidirectional stream with an explicit write queue:
        write() requests one at a time, close_writes() half-closes, and a
        status future resolves when the server finishes (the reference's
        client-streaming v2/v3 richness — client_streaming_v2.h:46,
        _v3.h:46: write queue + CloseWrites + status futures)."""
        return StreamingCall(self, service, method, resp_cls, on_response)

    def close(self):
        async def _close():
            await self.channel.close()

        try:
            asyncio.run_coroutine_threadsafe(_close(), self._loop).result(5)
        except Exception:
            pass
        self._loop.call_soon_threadsafe(self._loop.stop)
        self._thread.join(timeout=5)


class StreamingCall:
    """One live bidi stream over an AsyncClient (see open_stream)."""

    _CLOSE = object()

    def __init__(self, client: AsyncClient, service: str, method: str,
                 resp_cls, on_response=None):
        self._client = client
        self._resp_cls = resp_cls
        self._on_response = on_response
        self._responses: "queue.Queue" = queue.Queue()
        self.status: Future = Future()
        self._wq: Optional[asyncio.Queue] = None
        started = threading.Event()

        async def run():
            self._wq = asyncio.Queue()
            started.set()
            fn = client.channel.stream_stream(
                f"/{service}/{method}",
                request_serializer=lambda m: m.SerializeToString(),
                response_deserializer=resp_cls.FromString)

            async def gen():
                while True:
                    item = await self._wq.get()
                    if item is self._CLOSE:
                        return  # half-close: CloseWrites
                    yield item

            try:
                async for resp in fn(gen()):
                    self._responses.put(resp)
                    if self._on_response:
                        self._on_response(resp)
                self._responses.put(None)  # end-of-stream sentinel
                self.status.set_result(True)
            except BaseException as e:  # noqa: BLE001
                self._responses.put(None)
                self.status.set_exception(e)

        self._task = asyncio.run_coroutine_threadsafe(run(), client._loop)
        started.wait(timeout=10)

    def write(self, request) -> None:
        """Enqueue one request (returns immediately; ordered delivery)."""
        self._client._loop.call_soon_threadsafe(self._wq.put_nowait, request)

    def close_writes(self) -> None:
        """Half-close the write side; the server sees end-of-requests."""
        self._client._loop.call_soon_threadsafe(self._wq.put_nowait,
                                                self._CLOSE)

    def responses(self):
        """Iterate responses as they arrive (ends at server finish)."""
        while True:
            r = self._responses.get()
            if r is None:
                return
            yield r

    def cancel(self) -> None:
        self._task.cancel()


def siege(target: str, service: str, method: str, make_request: Callable[[int], Any],
          resp_cls, rate_hz: float = 100.0, duration_s: float = 5.0,
          max_outstanding: int = 950) -> dict:
    """Constant-rate load generator with an outstanding-request cap
    (reference siege.cc: max 950 outstanding, constant-rate issue loop).
    Returns latency/throughput stats."""
    import numpy as np

    client = AsyncClient(target)
    sem = threading.Semaphore(max_outstanding)
    lat: List[float] = []
    lock = threading.Lock()
    errors = [0]
    issued = 0
    t0 = time.monotonic()
    period = 1.0 / rate_hz
    futs = []
    while time.monotonic() - t0 < duration_s:
        target_t = t0 + issued * period
        now = time.monotonic()
        if now < target_t:
            time.sleep(target_t - now)
        sem.acquire()
        start = time.monotonic()
        f = client.call(service, method, make_request(issued), resp_cls,
                        timeout=30)

        def done(fut, start=start):
            with lock:
                if fut.exception():
                    errors[0] += 1
                else:
                    lat.append(time.monotonic() - start)
            sem.release()

        f.add_done_callback(done)
        futs.append(f)
        issued += 1
    for f in futs:
        try:
            f.result(timeout=60)
        except Exception:
            pass
    elapsed = time.monotonic() - t0
    client.close()
    lat_ms = np.array(sorted(lat)) * 1e3 if lat else np.array([0.0])
    return dict(issued=issued, completed=len(lat), errors=errors[0],
                seconds=elapsed, rate=len(lat) / elapsed,
                p50_ms=float(np.percentile(lat_ms, 50)),
                p90_ms=float(np.percentile(lat_ms, 90)),
                p99_ms=float(np.percentile(lat_ms, 99)))
