"""trtlab_amd.hip — device layer utilities (reference: trtlab/cuda).

DeviceInfo (reference device_info.h:35, NVML -> amd-smi/rocm-smi), stream
helpers, and device properties via the native module.
"""
from __future__ import annotations

import subprocess
from typing import Dict, List, Optional


def device_count() -> int:
    from trtlab_amd import native

    return native().hip.device_count()


def device_properties(device: int = 0) -> Dict:
    from trtlab_amd import native

    return native().hip.device_properties(device)


def synchronize() -> None:
    from trtlab_amd import native

    native().hip.device_synchronize()


class Stream:
    """RAII HIP stream (reference workspace.cc:12-18)."""

    def __init__(self):
        from trtlab_amd import native

        self._C = native()
        self.handle = self._C.hip.stream_create()

    def synchronize(self):
        self._C.hip.stream_synchronize(self.handle)

    def close(self):
        if self.handle:
            self._C.hip.stream_destroy(self.handle)
            self.handle = 0

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class DeviceInfo:
    """Power/memory introspection via rocm-smi (reference DeviceInfo uses
    NVML: device_info.cc:66-127)."""

    @staticmethod
    def power_usage(device: int = 0) -> Optional[float]:
        from trtlab_amd.utils.metrics import read_gpu_power

        return read_gpu_power(device)

    @staticmethod
    def memory_info(device: int = 0) -> Dict[str, int]:
        props = device_properties(device)
        return {"total": props["total_mem"]}

    @staticmethod
    def gcn_arch(device: int = 0) -> str:
        return device_properties(device)["gcn_arch"]

    @staticmethod
    def numa_node(device: int = 0) -> Optional[int]:
        """NUMA node of the GPU via sysfs (reference DeviceInfo::Affinity,
        device_info.cc:66 — NVML GPU<->CPU affinity -> /sys here)."""
        import glob

        cards = sorted(glob.glob("/sys/class/drm/card*/device/numa_node"))
        if device < len(cards):
            try:
                with open(cards[device]) as fh:
                    n = int(fh.read().strip())
                return n if n >= 0 else None
            except OSError:
                return None
        return None

    @staticmethod
    def cpu_affinity(device: int = 0) -> List[int]:
        """CPUs local to the GPU's NUMA node (for pinning the pre/post
        thread pools next to the device)."""
        node = DeviceInfo.numa_node(device)
        if node is None:
            return list(range(len(__import__("os").sched_getaffinity(0))))
        try:
            with open(f"/sys/devices/system/node/node{node}/cpulist") as fh:
                spec = fh.read().strip()
            cpus: List[int] = []
            for part in spec.split(","):
                if "-" in part:
                    a, b = part.split("-")
                    cpus.extend(range(int(a), int(b) + 1))
                elif part:
                    cpus.append(int(part))
            return cpus
        except OSError:
            return []
