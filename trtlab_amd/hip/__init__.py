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
