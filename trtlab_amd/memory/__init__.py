"""trtlab_amd.memory — the allocator layer.

MI355X redesign of the reference's foonathan-style allocator framework
(trtlab/memory: memory_type concept, block allocators, arenas, pools,
transactional stack, descriptors, tracking) plus trtlab/cuda's device/pinned
memory types. The GPU-facing primitives are native (csrc/runtime/memory.cpp
on hipMalloc/hipHostMalloc, 256-B aligned for HBM3E); this module adds the
host-side structure: RAII descriptors, blocking pools, transactional stacks
and the liveness-based arena planner the engine uses for activation reuse
(the contract the reference gets from TensorRT's
createExecutionContextWithoutDeviceMemory + setDeviceMemory,
trtlab/tensorrt/src/workspace.cc:40-41).
"""
from __future__ import annotations

import os

import threading

import numpy as np
from dataclasses import dataclass, field
from typing import List, Optional, Tuple

from trtlab_amd.utils import round_up

MIN_DEVICE_ALIGN = 256  # HBM3E-friendly minimum allocation alignment


# --------------------------------------------------------------------------
# Memory types (reference memory_type.h:93 — where do the bytes live)
class MemoryType:
    HOST = "host"
    PINNED = "host_pinned"
    DEVICE = "device"


# --------------------------------------------------------------------------
# Descriptors: RAII allocation handles (reference descriptor.h:40,102)
class DeviceBuffer:
    """Owns a hipMalloc'd region; frees on close()/del."""

    def __init__(self, nbytes: int, device: int = 0):
        from trtlab_amd import native

        self._C = native()
        self.nbytes = int(nbytes)
        self.device = device
        self.ptr = self._C.memory.device_malloc(self.nbytes, device)

    def close(self):
        if getattr(self, "ptr", 0):
            self._C.memory.device_free(self.ptr, self.nbytes)
            self.ptr = 0

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass

    def upload(self, buf) -> None:
        """Copy a host buffer (numpy array / bytes) to the device."""
        import numpy as np

        arr = np.ascontiguousarray(buf) if not isinstance(buf, (bytes, bytearray)) else np.frombuffer(buf, dtype=np.uint8)
        self._C.memory.memcpy_h2d(self.ptr, arr, min(arr.nbytes, self.nbytes))

    def download(self, arr) -> None:
        self._C.memory.memcpy_d2h(arr, self.ptr, min(arr.nbytes, self.nbytes))

    def dlpack(self, shape, dtype: str = "f16"):
        """Zero-copy DLPack capsule over this buffer (consume with
        torch.from_dlpack; reference core/types.h:83 DLPack interop).
        The buffer must outlive tensors created from the capsule."""
        code, bits = {"f16": (2, 16), "f32": (2, 32), "i32": (0, 32),
                      "u8": (1, 8), "bf16": (4, 16)}[dtype]
        return self._C.memory.to_dlpack(self.ptr, list(shape), code, bits,
                                        self.device)


class PinnedBuffer:
    """Page-locked host staging buffer (reference host_pinned_memory)."""

    def __init__(self, nbytes: int):
        from trtlab_amd import native

        self._C = native()
        self.nbytes = int(nbytes)
        self.ptr = self._C.memory.pinned_malloc(self.nbytes)

    def close(self):
        if getattr(self, "ptr", 0):
            self._C.memory.pinned_free(self.ptr, self.nbytes)
            self.ptr = 0

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class HugePageBuffer:
    """Huge-page-backed host buffer: explicit 2 MiB hugetlb pages when the
    system pool has them (`hugetlb` attr says which), transparent-huge-page
    madvise fallback otherwise. First-touch happens on the allocating
    thread — allocate from an affinity-pinned `ThreadPool` worker for NUMA
    locality. `pin=True` hipHostRegisters the range so the GPU can DMA it
    (requires a GPU). Reference: the huge-page raw allocator in
    trtlab/memory (memory/include/trtlab/memory/allocator/malloc.h family).
    """

    def __init__(self, nbytes: int, pin: bool = False):
        from trtlab_amd import native

        self._C = native()
        self.nbytes = int(nbytes)
        self.pin = bool(pin)
        self.ptr, self.hugetlb = self._C.memory.huge_malloc(self.nbytes,
                                                            self.pin)

    def array(self, dtype=np.uint8) -> np.ndarray:
        """Zero-copy numpy view of the buffer (valid while it lives)."""
        return np.frombuffer(
            self._C.memory.host_view(self.ptr, self.nbytes), dtype=dtype)

    def close(self):
        if getattr(self, "ptr", 0):
            self._C.memory.huge_free(self.ptr, self.nbytes, self.pin)
            self.ptr = 0

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def device_bytes_in_use() -> int:
    from trtlab_amd import native

    return native().memory.device_bytes_in_use()


# --------------------------------------------------------------------------
# Blocking block pool (reference pool.h v4 + cuda allocator pools):
# fixed-size device blocks, Pop blocks until a block is free, releases
# return the block — the concurrency-limiting primitive for Buffers.
class FirstTouchBuffer:
    """Host buffer whose pages are faulted in ("first-touched") from a
    thread pinned to a chosen NUMA node's CPUs, so the kernel places them
    in that node's memory — the reference's first_touch_allocator
    (core/memory/first_touch_allocator.h:35) for staging buffers feeding
    a GPU on a specific socket. Pure malloc + sched_setaffinity; pinned
    registration is the caller's choice (PinnedBuffer already faults via
    hipHostMalloc on the calling thread).

    node=-1 touches on the calling thread (plain first-touch)."""

    PAGE = 4096

    def __init__(self, nbytes: int, node: int = -1, topology=None):
        import ctypes
        import threading

        self.nbytes = int(nbytes)
        self._buf = (ctypes.c_char * self.nbytes)()
        self.ptr = ctypes.addressof(self._buf)
        self.node = node
        self.touched_on: List[int] = []

        def touch():
            if node >= 0:
                if topology is not None:
                    cpus = topology.nearest_cpus(node)
                else:
                    from trtlab_amd.core.numa import NumaTopology

                    cpus = NumaTopology().nearest_cpus(node)
                if cpus:
                    try:
                        os.sched_setaffinity(0, cpus)
                        self.touched_on = sorted(
                            os.sched_getaffinity(0))
                    except OSError:
                        pass  # restricted affinity mask (containers)
            mv = memoryview(self._buf).cast("B")
            for off in range(0, self.nbytes, self.PAGE):
                mv[off] = 0
            if self.nbytes:
                mv[-1] = 0

        if node >= 0:
            t = threading.Thread(target=touch, name="first-touch")
            t.start()
            t.join()
        else:
            touch()

    def close(self):
        self._buf = None
        self.ptr = 0


class BlockingBlockPool:
    def __init__(self, block_bytes: int, count: int, device: int = 0):
        from trtlab_amd import native

        self._native = native().memory.BlockPool(block_bytes, count, device)
        self._cv = threading.Condition()

    def pop(self, timeout: Optional[float] = None) -> int:
        import time as _time

        with self._cv:
            deadline = None if timeout is None else _time.monotonic() + timeout
            while True:
                p = self._native.acquire()
                if p:
                    return p
                # monotonic deadline: a notify that loses the race to another
                # waiter must not restart the full timeout window
                remaining = None
                if deadline is not None:
                    remaining = deadline - _time.monotonic()
                    if remaining <= 0:
                        raise TimeoutError("BlockingBlockPool.pop timed out")
                if not self._cv.wait(remaining):
                    raise TimeoutError("BlockingBlockPool.pop timed out")

    def push(self, ptr: int) -> None:
        with self._cv:
            self._native.release(ptr)
            self._cv.notify()

    @property
    def available(self) -> int:
        return self._native.available()

    @property
    def block_bytes(self) -> int:
        return self._native.block_bytes


# --------------------------------------------------------------------------
# Transactional stack (host-side mirror of transactional_allocator.h:156):
# LIFO request-scoped scratch offsets. Used for planning and host scratch.
class TransactionalStack:
    def __init__(self, capacity: int, alignment: int = MIN_DEVICE_ALIGN):
        self.capacity = capacity
        self.alignment = alignment
        self._top = 0
        self._marks: List[int] = []

    def allocate(self, nbytes: int) -> int:
        off = round_up(self._top, self.alignment)
        if off + nbytes > self.capacity:
            raise MemoryError(
                f"TransactionalStack overflow: {off + nbytes} > {self.capacity}"
            )
        self._top = off + nbytes
        return off

    def begin(self) -> None:
        self._marks.append(self._top)

    def commit(self) -> None:
        self._top = self._marks.pop()

    @property
    def high_water(self) -> int:
        return self._top


# --------------------------------------------------------------------------
# Liveness-based arena planner: assign byte offsets to tensors with
# [first_use, last_use] intervals so non-overlapping lifetimes share memory.
# This is the activation-arena sizing contract of the reference
# (inference_manager.cc:108-117 GetActivationsMemorySize) made explicit.
@dataclass
class _Interval:
    name: str
    size: int
    start: int
    end: int
    offset: int = -1


class ArenaPlanner:
    def __init__(self, alignment: int = MIN_DEVICE_ALIGN):
        self.alignment = alignment
        self._items: List[_Interval] = []

    def add(self, name: str, nbytes: int, first_use: int, last_use: int) -> None:
        self._items.append(
            _Interval(name, round_up(max(nbytes, 1), self.alignment), first_use, last_use)
        )

    def plan(self) -> Tuple[dict, int]:
        """Greedy best-fit by decreasing size. Returns ({name: offset}, total)."""
        items = sorted(self._items, key=lambda t: -t.size)
        placed: List[_Interval] = []
        total = 0
        for it in items:
            overlapping = [
                p for p in placed if not (p.end < it.start or p.start > it.end)
            ]
            overlapping.sort(key=lambda p: p.offset)
            best_off = None
            best_waste = None
            prev_end = 0
            for p in overlapping:
                gap = p.offset - prev_end
                if gap >= it.size:
                    waste = gap - it.size
                    if best_waste is None or waste < best_waste:
                        best_off, best_waste = prev_end, waste
                prev_end = max(prev_end, p.offset + p.size)
            if best_off is None:
                best_off = prev_end
            it.offset = best_off
            placed.append(it)
            total = max(total, it.offset + it.size)
        return {it.name: it.offset for it in items}, round_up(total, self.alignment)


# --------------------------------------------------------------------------
# Allocation-size histogram tracker (reference trackers.h:37
# histogram_tracker): wraps any allocate/deallocate pair with log2-bucket
# counts + byte counters; report() renders the reference-style table.
class HistogramTracker:
    def __init__(self, name: str = "alloc"):
        self.name = name
        self.buckets = [0] * 48      # count of allocations by ceil(log2)
        self.total_allocs = 0
        self.total_bytes = 0
        self.in_use = 0
        self.high_water = 0

    def on_allocate(self, nbytes: int) -> None:
        b = max(int(nbytes - 1).bit_length(), 0) if nbytes > 1 else 0
        self.buckets[min(b, 47)] += 1
        self.total_allocs += 1
        self.total_bytes += nbytes
        self.in_use += nbytes
        self.high_water = max(self.high_water, self.in_use)

    def on_deallocate(self, nbytes: int) -> None:
        self.in_use -= nbytes

    def report(self) -> str:
        lines = [f"[{self.name}] allocs={self.total_allocs} "
                 f"bytes={self.total_bytes} in_use={self.in_use} "
                 f"high_water={self.high_water}"]
        for i, c in enumerate(self.buckets):
            if c:
                lines.append(f"  2^{i:<2d} ({1 << i:>12d} B): {c}")
        return "\n".join(lines)


class TrackedDeviceAllocator:
    """Device allocator with a histogram tracker (reference
    make_tracked_allocator, tracking.h:370)."""

    def __init__(self, device: int = 0, name: str = "device"):
        from trtlab_amd import native

        self._C = native()
        self.device = device
        self.tracker = HistogramTracker(name)
        self._sizes: dict = {}

    def allocate(self, nbytes: int) -> int:
        p = self._C.memory.device_malloc(nbytes, self.device)
        self._sizes[p] = nbytes
        self.tracker.on_allocate(nbytes)
        return p

    def deallocate(self, ptr: int) -> None:
        nbytes = self._sizes.pop(ptr)
        self._C.memory.device_free(ptr, nbytes)
        self.tracker.on_deallocate(nbytes)


# --------------------------------------------------------------------------
# Debug fill/fence (reference debugging.h:1-114 + detail/debug_helpers.h:
# fill fresh memory with a pattern, fence both ends, verify on free).
FENCE_BYTES = 64
_FENCE = 0xFD
_FILL = 0xCD


class FencedHostBuffer:
    """Host buffer with guard fences: allocates nbytes + 2 fences, fills
    the payload with the debug pattern, and check()/close() raise on any
    fence corruption (buffer overrun detector for host staging code)."""

    def __init__(self, nbytes: int):
        self.nbytes = int(nbytes)
        self._raw = np.empty(self.nbytes + 2 * FENCE_BYTES, np.uint8)
        self._raw[:FENCE_BYTES] = _FENCE
        self._raw[-FENCE_BYTES:] = _FENCE
        self._raw[FENCE_BYTES:-FENCE_BYTES] = _FILL
        self.array = self._raw[FENCE_BYTES:FENCE_BYTES + self.nbytes]

    def check(self) -> None:
        if (self._raw[:FENCE_BYTES] != _FENCE).any():
            raise MemoryError("front fence corrupted (underrun)")
        if (self._raw[-FENCE_BYTES:] != _FENCE).any():
            raise MemoryError("back fence corrupted (overrun)")

    def close(self) -> None:
        self.check()


class FencedDeviceBuffer(DeviceBuffer):
    """DeviceBuffer with device-side guard fences (GPU debug builds)."""

    def __init__(self, nbytes: int, device: int = 0):
        super().__init__(int(nbytes) + 2 * FENCE_BYTES, device)
        self.payload = self.ptr + FENCE_BYTES
        self.payload_bytes = int(nbytes)
        pat = np.full(FENCE_BYTES, _FENCE, np.uint8)
        self._C.memory.memcpy_h2d(self.ptr, pat, FENCE_BYTES)
        self._C.memory.memcpy_h2d(self.payload + self.payload_bytes, pat,
                                  FENCE_BYTES)

    def check(self) -> None:
        got = np.zeros(FENCE_BYTES, np.uint8)
        self._C.memory.memcpy_d2h(got, self.ptr, FENCE_BYTES)
        if (got != _FENCE).any():
            raise MemoryError("front fence corrupted (underrun)")
        self._C.memory.memcpy_d2h(got, self.payload + self.payload_bytes,
                                  FENCE_BYTES)
        if (got != _FENCE).any():
            raise MemoryError("back fence corrupted (overrun)")


# --------------------------------------------------------------------------
# Composable allocator algebra (reference block_allocators.h:186-461:
# growing / count-limited / size-limited block allocators + factory).
# Wrappers compose over any inner allocator exposing allocate/deallocate —
# the native DeviceArena, TrackedDeviceAllocator, or a test fake.
class CountLimitedAllocator:
    """Cap the number of LIVE allocations (reference
    count_limited_block_allocator, block_allocators.h:258)."""

    def __init__(self, inner, max_count: int):
        self.inner = inner
        self.max_count = max_count
        self._live = 0

    def allocate(self, nbytes: int):
        if self._live >= self.max_count:
            raise MemoryError(
                f"allocation count limit reached ({self.max_count})")
        p = self.inner.allocate(nbytes)
        self._live += 1
        return p

    def deallocate(self, ptr) -> None:
        self.inner.deallocate(ptr)
        self._live = max(0, self._live - 1)


class SizeLimitedAllocator:
    """Cap the total LIVE bytes (reference size_limited_block_allocator,
    block_allocators.h:342). Sized for 288 GB of HBM3E by default."""

    def __init__(self, inner, max_bytes: int = 288 << 30):
        self.inner = inner
        self.max_bytes = max_bytes
        self._sizes: dict = {}
        self.in_use = 0

    def allocate(self, nbytes: int):
        if self.in_use + nbytes > self.max_bytes:
            raise MemoryError(
                f"size limit: {self.in_use} + {nbytes} > {self.max_bytes}")
        p = self.inner.allocate(nbytes)
        self._sizes[p] = nbytes
        self.in_use += nbytes
        return p

    def deallocate(self, ptr) -> None:
        self.inner.deallocate(ptr)
        self.in_use -= self._sizes.pop(ptr, 0)


class TrackedAllocator:
    """Attach a HistogramTracker to any allocator (reference
    make_tracked_allocator, tracking.h:370)."""

    def __init__(self, inner, name: str = "tracked"):
        self.inner = inner
        self.tracker = HistogramTracker(name)
        self._sizes: dict = {}

    def allocate(self, nbytes: int):
        p = self.inner.allocate(nbytes)
        self._sizes[p] = nbytes
        self.tracker.on_allocate(nbytes)
        return p

    def deallocate(self, ptr) -> None:
        self.inner.deallocate(ptr)
        self.tracker.on_deallocate(self._sizes.pop(ptr, 0))


def make_device_allocator(device: int = 0, max_bytes: int = 0,
                          max_count: int = 0, tracked: bool = False,
                          initial_bytes: int = 0):
    """Factory composing the device allocator stack (reference
    make_block_allocator, block_allocators.h:432): growing best-fit
    DeviceArena at the bottom, then optional size / count limits and
    tracking."""
    from trtlab_amd import native

    class _ArenaAdapter:
        def __init__(self):
            self.arena = native().memory.DeviceArena(device, initial_bytes,
                                                     max_bytes)

        def allocate(self, nbytes):
            return self.arena.allocate(nbytes)

        def deallocate(self, ptr):
            self.arena.deallocate(ptr)

    alloc = _ArenaAdapter()
    if max_bytes:
        alloc = SizeLimitedAllocator(alloc, max_bytes)
    if max_count:
        alloc = CountLimitedAllocator(alloc, max_count)
    if tracked:
        alloc = TrackedAllocator(alloc, f"device{device}")
    return alloc
