"""Plain cyclic buffer: a ring of fixed-size segments with per-segment
sync callbacks (reference trtlab/core/cyclic_buffer.h:1-239 +
src/cyclic_buffer.cc:1-266 — the non-windowed variant).

A producer appends bytes; when a segment fills, `on_segment(view, seq)`
fires (e.g. kick off an async H2D copy) and the segment is considered
in-flight until `release(seq)` — appending into a segment that has not
been released BLOCKS (back-pressure), mirroring the reference's sync
objects per segment.
"""
from __future__ import annotations

import threading
from typing import Callable, Optional

import numpy as np


class CyclicBuffer:
    def __init__(self, segment_bytes: int, segments: int,
                 on_segment: Optional[Callable[[memoryview, int], None]]
                 = None):
        if segment_bytes <= 0 or segments < 2:
            raise ValueError("need segment_bytes > 0 and >= 2 segments")
        self.segment_bytes = segment_bytes
        self.segments = segments
        self._buf = np.zeros(segment_bytes * segments, np.uint8)
        self._on_segment = on_segment
        self._cv = threading.Condition()
        self._inflight: set[int] = set()   # segment seq numbers not released
        self._seq = 0                      # next segment sequence number
        self._fill = 0                     # bytes in the current segment

    # ------------------------------------------------------------ producer
    def append(self, data, timeout: Optional[float] = None) -> int:
        """Append bytes; returns number of segments completed. Blocks when
        the ring wraps onto an unreleased segment (back-pressure)."""
        src = np.frombuffer(data, np.uint8) if not isinstance(
            data, np.ndarray) else data.view(np.uint8).reshape(-1)
        done = 0
        pos = 0
        while pos < len(src):
            slot = self._seq % self.segments
            with self._cv:
                # the previous occupant of this slot must have been released
                prev = self._seq - self.segments
                if not self._cv.wait_for(
                        lambda: prev < 0 or prev not in self._inflight,
                        timeout=timeout):
                    raise TimeoutError("cyclic buffer full (back-pressure)")
            take = min(len(src) - pos, self.segment_bytes - self._fill)
            base = slot * self.segment_bytes + self._fill
            self._buf[base:base + take] = src[pos:pos + take]
            self._fill += take
            pos += take
            if self._fill == self.segment_bytes:
                seq = self._seq
                with self._cv:
                    self._inflight.add(seq)
                    self._seq += 1
                    self._fill = 0
                if self._on_segment is not None:
                    view = memoryview(self._buf)[
                        slot * self.segment_bytes:
                        (slot + 1) * self.segment_bytes]
                    self._on_segment(view, seq)
                done += 1
        return done

    # ------------------------------------------------------------ consumer
    def release(self, seq: int) -> None:
        """Mark segment `seq` consumed (its memory may be overwritten)."""
        with self._cv:
            self._inflight.discard(seq)
            self._cv.notify_all()

    @property
    def inflight(self) -> int:
        return len(self._inflight)

    def segment_view(self, seq: int) -> memoryview:
        slot = seq % self.segments
        return memoryview(self._buf)[slot * self.segment_bytes:
                                     (slot + 1) * self.segment_bytes]
