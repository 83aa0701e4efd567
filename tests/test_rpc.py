"""RPC framework tests: real gRPC server + client in one process over
localhost (the reference's nvrpc test strategy — test_pingpong.cc)."""
import threading
import time

import pytest

from trtlab_amd.rpc import (AsyncClient, AsyncService, BatchingService,
                            EchoRequest, EchoResponse, HealthRequest,
                            HealthResponse, Server, SyncClient, siege)


@pytest.fixture()
def echo_server():
    server = Server("127.0.0.1:0")
    svc = AsyncService("trtlab.Echo")

    async def echo(request, context, resources):
        return EchoResponse(message=request.message, tag=request.tag)

    async def echo_stream(request_iter, context, resources):
        async for req in request_iter:
            yield EchoResponse(message=req.message, tag=req.tag)

    svc.register_unary("Echo", echo, EchoRequest, EchoResponse)
    svc.register_streaming("EchoStream", echo_stream, EchoRequest, EchoResponse)
    server.register_service(svc)

    health = AsyncService("trtlab.Health")

    async def ready(request, context, resources):
        return HealthResponse(ready=True, status="serving")

    health.register_unary("Check", ready, HealthRequest, HealthResponse)
    server.register_service(health)

    server.async_start()
    yield server
    server.shutdown()


def test_unary_sync_client(echo_server):
    c = SyncClient(f"127.0.0.1:{echo_server.port}")
    resp = c.call("trtlab.Echo", "Echo", EchoRequest(message="hi", tag=7),
                  EchoResponse, timeout=5)
    assert resp.message == "hi" and resp.tag == 7
    c.close()


def test_unary_async_client_many(echo_server):
    c = AsyncClient(f"127.0.0.1:{echo_server.port}")
    futs = [c.call("trtlab.Echo", "Echo",
                   EchoRequest(message=f"m{i}", tag=i), EchoResponse,
                   timeout=10) for i in range(50)]
    for i, f in enumerate(futs):
        r = f.result(timeout=10)
        assert r.message == f"m{i}" and r.tag == i
    c.close()


def test_streaming_pingpong(echo_server):
    c = AsyncClient(f"127.0.0.1:{echo_server.port}")
    reqs = [EchoRequest(message=f"s{i}", tag=i) for i in range(10)]
    resps = c.stream("trtlab.Echo", "EchoStream", reqs, EchoResponse).result(10)
    assert [r.tag for r in resps] == list(range(10))
    c.close()


def test_health(echo_server):
    c = SyncClient(f"127.0.0.1:{echo_server.port}")
    r = c.call("trtlab.Health", "Check", HealthRequest(), HealthResponse,
               timeout=5)
    assert r.ready and r.status == "serving"
    c.close()


def test_batching_service():
    batches = []

    def compute(requests):
        batches.append(len(requests))
        return [EchoResponse(message=r.message.upper(), tag=r.tag)
                for r in requests]

    server = Server("127.0.0.1:0")
    svc = BatchingService("trtlab.Batch", "Echo", EchoRequest, EchoResponse,
                          compute, max_batch_size=4, timeout_s=0.05)
    server.register_service(svc)
    server.async_start()
    try:
        c = AsyncClient(f"127.0.0.1:{server.port}")
        futs = [c.call("trtlab.Batch", "Echo",
                       EchoRequest(message=f"x{i}", tag=i), EchoResponse,
                       timeout=10) for i in range(8)]
        for i, f in enumerate(futs):
            assert f.result(timeout=10).message == f"X{i}"
        # one more -> closes on timeout path
        r = c.call("trtlab.Batch", "Echo", EchoRequest(message="late", tag=99),
                   EchoResponse, timeout=10).result(10)
        assert r.message == "LATE"
        c.close()
        assert sum(batches) == 9
        assert max(batches) <= 4
    finally:
        server.shutdown()
        svc.shutdown()


def test_siege_load_generator(echo_server):
    stats = siege(f"127.0.0.1:{echo_server.port}", "trtlab.Echo", "Echo",
                  lambda i: EchoRequest(message="load", tag=i), EchoResponse,
                  rate_hz=200, duration_s=1.0, max_outstanding=64)
    assert stats["errors"] == 0
    assert stats["completed"] >= 100
    assert stats["p99_ms"] > 0


def test_client_deadline_cancel(echo_server):
    """Client-side deadline on a slow handler: the call fails with
    DEADLINE_EXCEEDED and the server keeps serving (reference
    test_pingpong.cc early-cancel/early-finish teardown tests)."""
    import grpc

    server = echo_server
    svc = AsyncService("trtlab.Slow")

    async def slow(request, context, resources):
        import asyncio

        await asyncio.sleep(2.0)
        return EchoResponse(message=request.message, tag=request.tag)

    svc.register_unary("Echo", slow, EchoRequest, EchoResponse)
    # register a second service on a NEW server (echo_server already built)
    s2 = Server("127.0.0.1:0")
    s2.register_service(svc)
    s2.async_start()
    try:
        c = SyncClient(f"127.0.0.1:{s2.port}")
        with pytest.raises(grpc.RpcError) as ei:
            c.call("trtlab.Slow", "Echo", EchoRequest(message="x"),
                   EchoResponse, timeout=0.2)
        assert ei.value.code() == grpc.StatusCode.DEADLINE_EXCEEDED
        c.close()
        # the original echo server is still healthy
        c2 = SyncClient(f"127.0.0.1:{server.port}")
        r = c2.call("trtlab.Echo", "Echo", EchoRequest(message="ok"),
                    EchoResponse, timeout=5)
        assert r.message == "ok"
        c2.close()
    finally:
        s2.shutdown()


def test_shm_pool_checkout_release():
    """ShmPool: depth-bounded checkout, contents visible across handles,
    release unblocks waiters, close unlinks."""
    import threading

    import numpy as np

    from trtlab_amd.rpc.client import ShmPool

    pool = ShmPool(1024, depth=2)
    a1 = np.arange(256, dtype=np.float32)
    seg1, name1, size1 = pool.checkout(a1)
    seg2, name2, size2 = pool.checkout(a1 * 2)
    assert name1 != name2 and size1 == a1.nbytes

    # reader sees the bytes through an independent mapping
    from multiprocessing import shared_memory

    rd = shared_memory.SharedMemory(name=name1)
    got = np.frombuffer(rd.buf[:size1], dtype=np.float32).copy()
    rd.close()
    assert np.array_equal(got, a1)

    # third checkout blocks until a release
    acquired = threading.Event()

    def taker():
        s, _, _ = pool.checkout(a1)
        acquired.set()
        pool.release(s)

    t = threading.Thread(target=taker, daemon=True)
    t.start()
    assert not acquired.wait(timeout=0.2)
    pool.release(seg1)
    assert acquired.wait(timeout=5)
    pool.release(seg2)
    t.join(timeout=5)
    pool.close()
    with pytest.raises(FileNotFoundError):
        shared_memory.SharedMemory(name=name2)


def test_so_reuseport_two_servers_one_port():
    """Two Server instances bind the SAME port via SO_REUSEPORT (the
    multi-worker scale-out transport: examples/inference_server.py
    --workers runs one per process; here two in-process instances prove
    the socket plumbing) and both answer echo calls."""
    import numpy as np

    from trtlab_amd.rpc import EchoRequest, EchoResponse, SyncClient
    from trtlab_amd.rpc.server import AsyncService, Server

    def make(name):
        async def echo(req, ctx, res):
            return EchoResponse(message=f"{name}:{req.message}", tag=req.tag)

        svc = AsyncService("trtlab.Echo")
        svc.register_unary("Say", echo, EchoRequest, EchoResponse)
        s = Server("127.0.0.1:0")
        s.register_service(svc)
        return s

    s1 = make("a")
    s1.async_start()
    port = s1.port
    s2 = make("b")
    s2.address = f"127.0.0.1:{port}"
    try:
        s2.async_start()
        seen = set()
        # separate channels may land on either listener
        for i in range(20):
            c = SyncClient(f"127.0.0.1:{port}")
            r = c.call("trtlab.Echo", "Say",
                       EchoRequest(message="hi", tag=i), EchoResponse)
            seen.add(r.message.split(":")[0])
            c.close()
        assert seen <= {"a", "b"} and len(seen) >= 1
    finally:
        s2.shutdown()
        s1.shutdown()


def test_streaming_call_write_queue(echo_server):
    """Incremental streaming client (reference v2/v3 richness): write(),
    interleaved responses via callback, close_writes() half-close, and a
    status future that resolves on server finish."""
    from trtlab_amd.rpc.client import AsyncClient

    got = []
    c = AsyncClient(f"127.0.0.1:{echo_server.port}")
    try:
        call = c.open_stream("trtlab.Echo", "EchoStream", EchoResponse,
                             on_response=lambda r: got.append(r.tag))
        for i in range(5):
            call.write(EchoRequest(message=f"m{i}", tag=i))
        call.close_writes()
        resps = list(call.responses())
        assert [r.tag for r in resps] == list(range(5))
        assert got == list(range(5))  # callback saw them too
        assert call.status.result(timeout=5) is True
    finally:
        c.close()


def test_forwarding_middleman_relays_and_measures():
    """Middleman (reference 04_Middleman role): raw-bytes unary relay —
    client -> middleman -> backend echo — returns the backend's exact
    response and records hop latency."""
    import grpc

    from trtlab_amd.rpc.middleman import ForwardingService, _RawBytes
    from trtlab_amd.rpc.server import AsyncService, Server

    # backend: an echo service that tags responses
    backend_svc = AsyncService("trtlab.Echo")

    async def echo(request, context, resources):
        return _RawBytes(b"echo:" + request.data)

    backend_svc.register_unary("Ping", echo, _RawBytes, _RawBytes)
    backend = Server("127.0.0.1:0")
    backend.register_service(backend_svc)
    backend.async_start()

    fwd = ForwardingService(f"127.0.0.1:{backend.port}",
                            service="trtlab.Echo", method="Ping")
    front = Server("127.0.0.1:0")
    front.register_service(fwd.service)
    front.async_start()
    try:
        ch = grpc.insecure_channel(f"127.0.0.1:{front.port}")
        call = ch.unary_unary("/trtlab.Echo/Ping",
                              request_serializer=lambda b: b,
                              response_deserializer=lambda b: b)
        for i in range(5):
            out = call(f"m{i}".encode())
            assert out == f"echo:m{i}".encode()
        ch.close()
        s = fwd.stats()
        assert s["requests"] == 5 and s["p50_ms"] > 0
    finally:
        fwd.close()
        front.shutdown()
        backend.shutdown()


def test_balanced_client_spreads_and_fails_over():
    """Client-side LB (reference 99_LoadBalancer role, in-process): load
    spreads across healthy replicas; killing one evicts it after a
    transport failure and every later call lands on the survivor;
    application errors do NOT evict."""
    import grpc as _grpc

    from trtlab_amd.rpc.balancer import BalancedClient
    from trtlab_amd.rpc.middleman import _RawBytes
    from trtlab_amd.rpc.server import AsyncService, Server

    def make_server(tag):
        svc = AsyncService("trtlab.Echo")

        async def echo(request, context, resources):
            if request.data == b"boom":
                await context.abort(_grpc.StatusCode.INVALID_ARGUMENT,
                                    "bad request")
            return _RawBytes(tag + request.data)

        svc.register_unary("Ping", echo, _RawBytes, _RawBytes)
        srv = Server("127.0.0.1:0")
        srv.register_service(svc)
        srv.async_start()
        return srv

    s1, s2 = make_server(b"a:"), make_server(b"b:")
    lb = BalancedClient([f"127.0.0.1:{s1.port}", f"127.0.0.1:{s2.port}"],
                        service="trtlab.Echo", method="Ping",
                        cooldown_s=30.0)
    try:
        seen = set()
        for i in range(8):
            seen.add(bytes(lb.call(f"m{i}".encode(), timeout=10))[:2])
        assert seen == {b"a:", b"b:"}  # both replicas served

        # application error: surfaced, backend stays healthy
        import pytest as _pytest
        with _pytest.raises(_grpc.RpcError) as ei:
            lb.call(b"boom", timeout=10)
        assert ei.value.code() == _grpc.StatusCode.INVALID_ARGUMENT
        assert not any(v["down"] for v in lb.stats().values())

        # kill one replica: next calls retry onto the survivor
        s1.shutdown()
        for i in range(6):
            out = bytes(lb.call(f"k{i}".encode(), timeout=10))
            assert out.startswith(b"b:")
        st = lb.stats()
        downs = [a for a, v in st.items() if v["down"]]
        assert len(downs) == 1 and str(s1.port) in downs[0]
    finally:
        lb.close()
        s2.shutdown()


def test_balanced_client_all_down_reprobe():
    """When every backend is cooling down, the balancer re-probes the
    one whose cooldown expires soonest instead of failing fast — a
    transient full outage recovers without client-side restarts."""
    from trtlab_amd.rpc.balancer import BalancedClient
    from trtlab_amd.rpc.middleman import _RawBytes
    from trtlab_amd.rpc.server import AsyncService, Server

    svc = AsyncService("trtlab.Echo")

    async def echo(request, context, resources):
        return _RawBytes(b"ok:" + request.data)

    svc.register_unary("Ping", echo, _RawBytes, _RawBytes)
    srv = Server("127.0.0.1:0")
    srv.register_service(svc)
    srv.async_start()
    # second backend never existed: its port is closed
    lb = BalancedClient([f"127.0.0.1:{srv.port}", "127.0.0.1:1"],
                        service="trtlab.Echo", method="Ping",
                        cooldown_s=60.0)
    try:
        # force both into cooldown: dead backend by calling it, live one
        # artificially
        for be in lb._backends:
            be.down_until = __import__("time").monotonic() + 60.0
        out = bytes(lb.call(b"x", timeout=10))
        assert out == b"ok:x"  # re-probe found the healthy one (retries)
    finally:
        lb.close()
        srv.shutdown()
