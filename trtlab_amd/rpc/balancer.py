"""Client-side load balancer: least-outstanding routing over N server
addresses with failure-driven eviction and cooldown re-probe.

In-process counterpart of the reference's envoy L7 recipe
(99_LoadBalancer, ~150 us/txn overhead measured there) — when the
client can hold the replica list itself, the extra hop disappears and
failover decisions use the caller's own error observations. Pairs with
the server-side ReplicaGroup (parallel/__init__.py) which balances
execution contexts behind ONE endpoint; this balances across endpoints.
"""
from __future__ import annotations

import threading
import time
from typing import Callable, List, Optional, Sequence

import grpc


class _Backend:
    def __init__(self, address: str):
        self.address = address
        self.channel = grpc.insecure_channel(address)
        self.outstanding = 0
        self.failures = 0
        self.down_until = 0.0  # monotonic deadline of the cooldown
        self.calls = 0


class BalancedClient:
    """Unary client over a replica set.

    call(request) picks the healthy backend with the fewest outstanding
    requests; transport failures (UNAVAILABLE / DEADLINE_EXCEEDED) evict
    the backend for `cooldown_s` and the request retries on the next
    backend. Serializers default to raw bytes (any unary method can be
    balanced without its message classes, like the middleman relay)."""

    def __init__(self, addresses: Sequence[str],
                 service: str = "trtlab.Inference", method: str = "Infer",
                 request_serializer: Optional[Callable] = None,
                 response_deserializer: Optional[Callable] = None,
                 cooldown_s: float = 2.0, retries: int = 3):
        if not addresses:
            raise ValueError("need at least one backend address")
        self._mu = threading.Lock()
        self._cooldown = cooldown_s
        self._retries = retries
        ser = request_serializer or (lambda b: b)
        de = response_deserializer or (lambda b: b)
        self._backends: List[_Backend] = []
        self._calls = []
        for a in addresses:
            be = _Backend(a)
            self._backends.append(be)
            self._calls.append(be.channel.unary_unary(
                f"/{service}/{method}", request_serializer=ser,
                response_deserializer=de))

    def _pick(self) -> Optional[int]:
        now = time.monotonic()
        best, best_key = None, None
        with self._mu:
            for i, be in enumerate(self._backends):
                if be.down_until > now:
                    continue
                # least outstanding; ties broken by total calls so
                # sequential (never-concurrent) traffic still spreads
                key = (be.outstanding, be.calls)
                if best is None or key < best_key:
                    best, best_key = i, key
            if best is not None:
                self._backends[best].outstanding += 1
                self._backends[best].calls += 1
        return best

    def _done(self, i: int, ok: bool) -> None:
        with self._mu:
            be = self._backends[i]
            be.outstanding -= 1
            if ok:
                be.failures = 0
            else:
                be.failures += 1
                be.down_until = time.monotonic() + self._cooldown
    def call(self, request, timeout: float = 30.0):
        last_err: Optional[Exception] = None
        for _ in range(self._retries):
            i = self._pick()
            if i is None:  # every backend cooling down: probe the oldest
                with self._mu:
                    i = min(range(len(self._backends)),
                            key=lambda j: self._backends[j].down_until)
                    self._backends[i].down_until = 0.0
                    self._backends[i].outstanding += 1
                    self._backends[i].calls += 1
            try:
                resp = self._calls[i](request, timeout=timeout)
                self._done(i, True)
                return resp
            except grpc.RpcError as e:
                code = e.code()
                transport = code in (grpc.StatusCode.UNAVAILABLE,
                                     grpc.StatusCode.DEADLINE_EXCEEDED)
                self._done(i, not transport)
                if not transport:
                    raise  # application error: the backend is healthy
                last_err = e
        raise last_err  # all retries were transport failures

    def stats(self) -> dict:
        now = time.monotonic()
        with self._mu:
            return {be.address: dict(calls=be.calls,
                                     outstanding=be.outstanding,
                                     down=be.down_until > now)
                    for be in self._backends}

    def close(self) -> None:
        for be in self._backends:
            be.channel.close()
