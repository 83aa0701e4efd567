"""Cyclic windowed buffer: sliding windows with overlap over an unbounded
stream (reference: trtlab/core cyclic_windowed_buffer.h:59-441 + the GPU
specialization cuda/cyclic_windowed_buffer.h:11-44 — audio/streaming-style
input for inference: each compute window shares `overlap` trailing samples
with its predecessor; wrap-around replicates the overlap region so every
window is contiguous).
"""
from __future__ import annotations

import threading
from typing import Callable, Iterator, List, Optional

import numpy as np


class CyclicWindowedBuffer:
    """Host-side windowing logic (window_size, overlap in samples).

    push(data) appends samples; ready windows are emitted via the callback
    or collected with pop_windows(). A window is [n-overlap shared | new].
    """

    def __init__(self, window_size: int, overlap: int,
                 capacity_windows: int = 16, sample_shape=(),
                 dtype=np.float32,
                 on_window: Optional[Callable[[np.ndarray, int], None]] = None):
        if not 0 <= overlap < window_size:
            raise ValueError("need 0 <= overlap < window_size")
        self.window_size = window_size
        self.overlap = overlap
        self.stride = window_size - overlap
        self.sample_shape = tuple(sample_shape)
        self.dtype = dtype
        self._on_window = on_window
        # ring of full windows; the overlap is replicated on wrap (same
        # memory behavior as the reference's reserved stack, so a window is
        # always one contiguous slice)
        self._capacity = capacity_windows
        self._pending = np.zeros((0, *self.sample_shape), dtype)
        self._window_id = 0
        self._out: List[np.ndarray] = []

    def push(self, data: np.ndarray) -> int:
        """Append samples; returns number of windows emitted."""
        data = np.asarray(data, self.dtype)
        if data.shape[1:] != self.sample_shape:
            raise ValueError(f"sample shape {data.shape[1:]} != {self.sample_shape}")
        self._pending = np.concatenate([self._pending, data], axis=0)
        emitted = 0
        while len(self._pending) >= self.window_size:
            win = self._pending[:self.window_size].copy()
            if self._on_window:
                self._on_window(win, self._window_id)
            else:
                self._out.append(win)
            self._window_id += 1
            emitted += 1
            # keep the trailing overlap for the next window
            self._pending = self._pending[self.stride:]
        return emitted

    def pop_windows(self) -> List[np.ndarray]:
        out, self._out = self._out, []
        return out

    @property
    def windows_emitted(self) -> int:
        return self._window_id


class DeviceCyclicWindowedStack:
    """GPU variant: windows are assembled in pinned host memory and copied
    to a device ring; `on_compute_window` receives the device pointer of a
    contiguous window (reference cuda/cyclic_windowed_buffer.h:11-44 — copy
    + replicate via hipMemcpyAsync)."""

    def __init__(self, window_size: int, overlap: int, sample_bytes: int,
                 capacity_windows: int = 8, device: int = 0):
        from trtlab_amd import native

        self._C = native()
        self.window_bytes = window_size * sample_bytes
        self.window_size = window_size
        self.overlap = overlap
        self.capacity = capacity_windows
        self._dev = self._C.memory.device_malloc(
            self.window_bytes * capacity_windows, device)
        self._slot = 0

    def stage_window(self, host_window: np.ndarray) -> int:
        """Copy one contiguous window to the device ring; returns the device
        pointer of the staged window."""
        assert host_window.nbytes == self.window_bytes
        ptr = self._dev + self._slot * self.window_bytes
        self._C.memory.memcpy_h2d(ptr, np.ascontiguousarray(host_window),
                                  self.window_bytes)
        self._slot = (self._slot + 1) % self.capacity
        return ptr

    def close(self):
        if self._dev:
            self._C.memory.device_free(self._dev,
                                       self.window_bytes * self.capacity)
            self._dev = 0

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class ReservedWindowedStack(CyclicWindowedBuffer):
    """Zero-copy producer variant (reference
    cyclic_windowed_reserved_stack::reserve_window,
    cyclic_windowed_buffer.h:287-365): reserve_window() hands the producer
    a writable view of the NEXT window's fresh region — the overlap with
    the previous window is pre-filled — and commit_window() emits it.
    Avoids the push() concatenate/copy for producers that can write in
    place (e.g. a decoder writing feature frames)."""

    def __init__(self, window_size: int, overlap: int, sample_shape=(),
                 dtype=np.float32, on_window=None):
        super().__init__(window_size, overlap, sample_shape=sample_shape,
                         dtype=dtype, on_window=on_window)
        self._reserved: Optional[np.ndarray] = None

    def reserve_window(self) -> np.ndarray:
        """Writable [stride, *sample_shape] view for the window's NEW
        samples (the first `overlap` samples are carried over)."""
        if self._reserved is not None:
            raise RuntimeError("a window is already reserved (commit first)")
        self._win = np.zeros((self.window_size, *self.sample_shape),
                             self.dtype)
        if self.overlap and len(self._pending) >= self.overlap:
            self._win[:self.overlap] = self._pending[-self.overlap:]
        self._reserved = self._win[self.overlap:]
        return self._reserved

    def commit_window(self) -> int:
        """Emit the reserved window; returns its window id."""
        if self._reserved is None:
            raise RuntimeError("no reserved window")
        wid = self._window_id
        if self._on_window:
            self._on_window(self._win, wid)
        else:
            self._out.append(self._win)
        self._window_id += 1
        # trailing overlap becomes the next window's carried samples
        self._pending = self._win[-self.overlap:].copy() if self.overlap \
            else np.zeros((0, *self.sample_shape), self.dtype)
        self._reserved = None
        return wid


class WindowedTaskExecutor:
    """Per-window compute executor (reference
    cyclic_windowed_task_executor::on_compute_window,
    cyclic_windowed_buffer.h:369-440): push samples in, `compute_fn`
    runs on a worker thread for every completed window, results are
    collected in order."""

    def __init__(self, window_size: int, overlap: int, compute_fn,
                 sample_shape=(), dtype=np.float32, workers: int = 1):
        from trtlab_amd.core import ThreadPool

        self._pool = ThreadPool(workers, "windowed")
        self._results: dict = {}
        self._lock = threading.Lock()
        self._futs: List = []

        def on_window(win: np.ndarray, wid: int):
            def run(w=win.copy(), i=wid):
                r = compute_fn(w, i)
                with self._lock:
                    self._results[i] = r

            self._futs.append(self._pool.enqueue(run))

        self.buffer = CyclicWindowedBuffer(window_size, overlap,
                                           sample_shape=sample_shape,
                                           dtype=dtype, on_window=on_window)

    def push(self, data: np.ndarray) -> int:
        return self.buffer.push(data)

    def results(self) -> List:
        """Wait for all scheduled windows; results in window order."""
        for f in self._futs:
            f.result()
        with self._lock:
            return [self._results[i] for i in sorted(self._results)]

    def shutdown(self):
        self._pool.shutdown()
