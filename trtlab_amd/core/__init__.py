"""trtlab_amd.core — host concurrency & batching primitives.

MI355X redesign of trtlab/core: ThreadPool (thread_pool.h:87), blocking
resource Pool with return-to-pool semantics (pool.h:456 v4), StandardBatcher
(batcher.h:24), Dispatcher (dispatcher.h:30) and DeferredShortTaskPool
(task_pool.h:36). Pure host code, CPU-testable.
"""
from __future__ import annotations

import heapq
import os
import queue
import threading
import time
from concurrent.futures import Future
from dataclasses import dataclass, field
from typing import Any, Callable, Generic, List, Optional, Sequence, TypeVar

T = TypeVar("T")


# --------------------------------------------------------------------------
class ThreadPool:
    """Work-queue thread pool with optional per-thread CPU affinity
    (reference thread_pool.h:87, CreateThread pins affinity :256-274)."""

    def __init__(self, workers: int, name: str = "pool",
                 cpus: Optional[Sequence[int]] = None):
        self._q: "queue.Queue" = queue.Queue()
        self._threads: List[threading.Thread] = []
        self._shutdown = False
        self.name = name
        for i in range(workers):
            t = threading.Thread(
                target=self._run, args=(cpus[i % len(cpus)] if cpus else None,),
                name=f"{name}-{i}", daemon=True)
            t.start()
            self._threads.append(t)

    def _run(self, cpu: Optional[int]):
        if cpu is not None and hasattr(os, "sched_setaffinity"):
            try:
                os.sched_setaffinity(0, {cpu})
            except OSError:
                pass
        while True:
            item = self._q.get()
            if item is None:
                return
            fn, args, kwargs, fut = item
            if fut.set_running_or_notify_cancel():
                try:
                    fut.set_result(fn(*args, **kwargs))
                except BaseException as e:  # noqa: BLE001
                    fut.set_exception(e)

    def enqueue(self, fn: Callable, *args, **kwargs) -> Future:
        if self._shutdown:
            raise RuntimeError("ThreadPool is shut down")
        fut: Future = Future()
        self._q.put((fn, args, kwargs, fut))
        return fut

    def size(self) -> int:
        return len(self._threads)

    def shutdown(self):
        self._shutdown = True
        for _ in self._threads:
            self._q.put(None)
        for t in self._threads:
            t.join(timeout=5)


# --------------------------------------------------------------------------
class Pool(Generic[T]):
    """Blocking resource pool. `pop()` returns a checkout whose release
    (context-manager exit or .release()) returns the item to the pool —
    the concurrency-limiting primitive for execution contexts and buffers
    (reference pool.h:456 pop_unique/pop_shared :520-535)."""

    class Checkout(Generic[T]):
        __slots__ = ("item", "_pool", "_on_return", "_released")

        def __init__(self, item: T, pool: "Pool[T]", on_return=None):
            self.item = item
            self._pool = pool
            self._on_return = on_return
            self._released = False

        def release(self):
            if not self._released:
                self._released = True
                if self._on_return:
                    self._on_return(self.item)
                self._pool._push(self.item)

        def __enter__(self) -> T:
            return self.item

        def __exit__(self, *exc):
            self.release()
            return False

        def __del__(self):
            try:
                self.release()
            except Exception:
                pass

    def __init__(self, items: Sequence[T] = ()):
        self._q: "queue.Queue[T]" = queue.Queue()
        self._count = 0
        for it in items:
            self.push(it)

    @classmethod
    def create(cls, factory: Callable[[], T], count: int) -> "Pool[T]":
        return cls([factory() for _ in range(count)])

    def push(self, item: T) -> None:
        self._count += 1
        self._q.put(item)

    def _push(self, item: T) -> None:
        self._q.put(item)

    def pop(self, timeout: Optional[float] = None, on_return=None) -> "Pool.Checkout[T]":
        try:
            item = self._q.get(timeout=timeout)
        except queue.Empty:
            raise TimeoutError("Pool.pop timed out") from None
        return Pool.Checkout(item, self, on_return)

    @property
    def size(self) -> int:
        return self._count

    @property
    def available(self) -> int:
        return self._q.qsize()


# --------------------------------------------------------------------------
@dataclass
class Batch:
    items: List[Any] = field(default_factory=list)
    futures: List[Future] = field(default_factory=list)
    batch_id: int = 0
    created: float = 0.0
    deadline: float = 0.0


class StandardBatcher:
    """Batching *logic only* — no locks, no threads (reference batcher.h:24:
    enqueue :101, update -> closes on max_batch_size, close_batch on
    timeout). The Dispatcher provides concurrency around it."""

    def __init__(self, max_batch_size: int, timeout_s: float):
        self.max_batch_size = max_batch_size
        self.timeout_s = timeout_s
        self._next_id = 0
        self._open: Optional[Batch] = None

    def enqueue(self, item: Any) -> tuple[Future, Optional[Batch]]:
        """Returns (future, closed_batch_or_None)."""
        fut: Future = Future()
        if self._open is None:
            now = time.monotonic()
            self._open = Batch(batch_id=self._next_id, created=now,
                               deadline=now + self.timeout_s)
            self._next_id += 1
        b = self._open
        b.items.append(item)
        b.futures.append(fut)
        closed = None
        if len(b.items) >= self.max_batch_size:
            closed = self.close_batch()
        return fut, closed

    def close_batch(self) -> Optional[Batch]:
        b, self._open = self._open, None
        return b

    @property
    def open_batch(self) -> Optional[Batch]:
        return self._open


class DeferredShortTaskPool:
    """Single thread executing deadline-ordered short tasks (reference
    task_pool.h:36) — implements the batching-window timeout."""

    def __init__(self):
        self._heap: List[tuple[float, int, Callable]] = []
        self._cv = threading.Condition()
        self._seq = 0
        self._shutdown = False
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="deferred-tasks")
        self._thread.start()

    def enqueue_deferred(self, deadline: float, fn: Callable) -> None:
        with self._cv:
            heapq.heappush(self._heap, (deadline, self._seq, fn))
            self._seq += 1
            self._cv.notify()

    def _run(self):
        while True:
            with self._cv:
                while not self._heap and not self._shutdown:
                    self._cv.wait()
                if self._shutdown:
                    return
                deadline, _, fn = self._heap[0]
                now = time.monotonic()
                if deadline > now:
                    self._cv.wait(deadline - now)
                    continue
                heapq.heappop(self._heap)
            try:
                fn()
            except Exception:  # pragma: no cover — log, keep the timer alive
                import traceback

                traceback.print_exc()

    def shutdown(self):
        with self._cv:
            self._shutdown = True
            self._cv.notify()
        self._thread.join(timeout=5)


class Dispatcher:
    """Pairs a StandardBatcher with execution + timeout machinery
    (reference dispatcher.h:30): worker pool runs compute_batch_fn, a
    deferred task closes the window on timeout."""

    def __init__(self, max_batch_size: int, timeout_s: float,
                 compute_batch_fn: Callable[[List[Any]], List[Any]],
                 workers: int = 1):
        self._batcher = StandardBatcher(max_batch_size, timeout_s)
        self._fn = compute_batch_fn
        self._pool = ThreadPool(workers, name="dispatch")
        self._timers = DeferredShortTaskPool()
        self._mu = threading.Lock()

    def enqueue(self, item: Any) -> Future:
        with self._mu:
            fut, closed = self._batcher.enqueue(item)
            if closed is None and len(self._batcher.open_batch.items) == 1:
                bid = self._batcher.open_batch.batch_id
                self._timers.enqueue_deferred(
                    self._batcher.open_batch.deadline,
                    lambda: self._close_if_open(bid))
        if closed:
            self._submit(closed)
        return fut

    def _close_if_open(self, batch_id: int):
        with self._mu:
            b = self._batcher.open_batch
            closed = self._batcher.close_batch() if (b and b.batch_id == batch_id) else None
        if closed:
            self._submit(closed)

    def _submit(self, batch: Batch):
        def run():
            try:
                results = self._fn(batch.items)
                for f, r in zip(batch.futures, results):
                    f.set_result(r)
            except BaseException as e:  # noqa: BLE001
                for f in batch.futures:
                    if not f.done():
                        f.set_exception(e)

        self._pool.enqueue(run)

    def shutdown(self):
        self._pool.shutdown()
        self._timers.shutdown()


# --------------------------------------------------------------------------
class Resources:
    """Dependency-injection base for RPC contexts (reference resources.h:33)."""
    pass
