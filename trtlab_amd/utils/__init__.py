"""trtlab_amd.utils — shared helpers (reference: trtlab/core utils.h:31-42
BytesToString/StringToBytes, glog-style logging)."""
from __future__ import annotations

import logging
import re

log = logging.getLogger("trtlab_amd")

_UNITS = {
    "": 1,
    "b": 1,
    "kb": 1000,
    "mb": 1000**2,
    "gb": 1000**3,
    "tb": 1000**4,
    "kib": 1024,
    "mib": 1024**2,
    "gib": 1024**3,
    "tib": 1024**4,
}


def string_to_bytes(s: str) -> int:
    """Parse human byte sizes: '10MiB' -> 10485760 (reference utils.cc:60)."""
    m = re.fullmatch(r"\s*([0-9]*\.?[0-9]+)\s*([A-Za-z]*)\s*", str(s))
    if not m:
        raise ValueError(f"cannot parse byte size: {s!r}")
    val, unit = float(m.group(1)), m.group(2).lower()
    if unit not in _UNITS:
        raise ValueError(f"unknown byte unit {unit!r} in {s!r}")
    return int(val * _UNITS[unit])


def bytes_to_string(n: int) -> str:
    """Format bytes with binary units (reference utils.cc:44)."""
    n = float(n)
    for unit in ("B", "KiB", "MiB", "GiB", "TiB"):
        if abs(n) < 1024.0 or unit == "TiB":
            return f"{n:.1f} {unit}" if unit != "B" else f"{int(n)} B"
        n /= 1024.0
    return f"{n:.1f} TiB"


def round_up(a: int, b: int) -> int:
    return ((a + b - 1) // b) * b
