"""Prometheus metrics for the inference service (reference: 02_TensorRT_GRPC
src/metrics.cc:34-50 Exposer + server.cc:86-107 compute/request summaries,
load-ratio histogram, power gauge via NVML -> here amd-smi / rocm-smi)."""
from __future__ import annotations

import subprocess
from typing import Optional

from prometheus_client import (CollectorRegistry, Gauge, Histogram,
                               start_http_server)

_LOAD_RATIO_BUCKETS = (1.25, 1.5, 2.0, 10.0, 100.0)  # reference server.cc:96


class Metrics:
    _instance: Optional["Metrics"] = None

    def __init__(self, port: int = 50078):
        # own registry: instantiable more than once per process (tests,
        # multiple servers)
        self.registry = CollectorRegistry()
        self.compute_ms = Histogram(
            "trtlab_compute_duration_ms", "GPU compute duration per request",
            buckets=(1, 2, 3, 5, 8, 13, 21, 34, 55, 89),
            registry=self.registry)
        self.request_ms = Histogram(
            "trtlab_request_duration_ms", "total request duration",
            buckets=(1, 2, 3, 5, 8, 13, 21, 34, 55, 89, 144, 233),
            registry=self.registry)
        self.load_ratio = Histogram(
            "trtlab_load_ratio", "request/compute duration ratio "
            "(queueing pressure — the reference's autoscaling signal)",
            buckets=_LOAD_RATIO_BUCKETS, registry=self.registry)
        self.power_w = Gauge("trtlab_gpu_power_watts", "GPU power draw",
                             registry=self.registry)
        # shared device-arena pool gauges (native DeviceArena stats —
        # reference histogram_tracker / memory tracking role)
        self.arena_in_use = Gauge("trtlab_arena_in_use_bytes",
                                  "shared device arena bytes in use",
                                  registry=self.registry)
        self.arena_high_water = Gauge("trtlab_arena_high_water_bytes",
                                      "shared device arena high-water mark",
                                      registry=self.registry)
        self.arena_capacity = Gauge("trtlab_arena_capacity_bytes",
                                    "shared device arena capacity",
                                    registry=self.registry)
        self.port = port
        self._started = False

    @classmethod
    def initialize(cls, port: int = 50078) -> "Metrics":
        if cls._instance is None:
            cls._instance = cls(port)
            start_http_server(port, registry=cls._instance.registry)
            cls._instance._started = True
        return cls._instance

    def observe(self, compute_ms: float, request_ms: float) -> None:
        self.compute_ms.observe(compute_ms)
        self.request_ms.observe(request_ms)
        if compute_ms > 0:
            self.load_ratio.observe(request_ms / compute_ms)

    def update_arena(self, manager) -> None:
        """Refresh pool gauges from an InferenceManager's shared arena."""
        s = manager.arena_stats() if manager is not None else None
        if s:
            self.arena_in_use.set(s["in_use"])
            self.arena_high_water.set(s["high_water"])
            self.arena_capacity.set(s["capacity"])

    def update_power(self, device: int = 0) -> Optional[float]:
        """Refresh the power gauge from rocm-smi/amd-smi (reference:
        Server::Run's 2 s NVML control lambda, server.cc:322-330)."""
        w = read_gpu_power(device)
        if w is not None:
            self.power_w.set(w)
        return w


def read_gpu_power(device: int = 0) -> Optional[float]:
    try:
        out = subprocess.run(
            ["rocm-smi", "-d", str(device), "--showpower", "--json"],
            capture_output=True, text=True, timeout=5)
        if out.returncode == 0:
            import json

            data = json.loads(out.stdout)
            for card in data.values():
                for k, v in card.items():
                    if "Power" in k:
                        return float(v)
    except Exception:
        pass
    return None
