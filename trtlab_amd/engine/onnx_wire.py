"""Minimal protobuf wire codec for ONNX (no `onnx` package in the offline
image; protobuf wire format is stable and simple). Only the fields the
importer/exporter need. Field numbers from onnx.proto3 (public schema).

Wire types: 0 varint, 1 fixed64, 2 length-delimited, 5 fixed32.
"""
from __future__ import annotations

import struct
from typing import Dict, Iterator, List, Tuple


# ------------------------------------------------------------------ decode
def _read_varint(buf: bytes, i: int) -> Tuple[int, int]:
    result = 0
    shift = 0
    while True:
        b = buf[i]
        i += 1
        result |= (b & 0x7F) << shift
        if not (b & 0x80):
            return result, i
        shift += 7


def iter_fields(buf: bytes) -> Iterator[Tuple[int, int, object]]:
    """Yield (field_number, wire_type, value). Length-delimited values are
    bytes; varints are ints."""
    i = 0
    n = len(buf)
    while i < n:
        tag, i = _read_varint(buf, i)
        field, wt = tag >> 3, tag & 7
        if wt == 0:
            v, i = _read_varint(buf, i)
        elif wt == 1:
            v = struct.unpack_from("<q", buf, i)[0]
            i += 8
        elif wt == 2:
            ln, i = _read_varint(buf, i)
            v = buf[i:i + ln]
            i += ln
        elif wt == 5:
            v = struct.unpack_from("<i", buf, i)[0]
            i += 4
        else:
            raise ValueError(f"unsupported wire type {wt}")
        yield field, wt, v


def fields_dict(buf: bytes) -> Dict[int, List[object]]:
    d: Dict[int, List[object]] = {}
    for f, _, v in iter_fields(buf):
        d.setdefault(f, []).append(v)
    return d


def zigzag_to_int(v: int) -> int:
    return (v >> 1) ^ -(v & 1)


def varint_to_sint64(v: int) -> int:
    """Protobuf int64 on the wire is two's-complement in 64 bits."""
    return v - (1 << 64) if v >= (1 << 63) else v


def decode_packed_varints(buf: bytes) -> List[int]:
    out = []
    i = 0
    while i < len(buf):
        v, i = _read_varint(buf, i)
        out.append(varint_to_sint64(v))
    return out


# ------------------------------------------------------------------ encode
def _varint(v: int) -> bytes:
    if v < 0:
        v += 1 << 64
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def tag(field: int, wt: int) -> bytes:
    return _varint((field << 3) | wt)


def f_varint(field: int, v: int) -> bytes:
    return tag(field, 0) + _varint(v)


def f_bytes(field: int, v: bytes) -> bytes:
    return tag(field, 2) + _varint(len(v)) + v


def f_string(field: int, s: str) -> bytes:
    return f_bytes(field, s.encode())


def f_float(field: int, v: float) -> bytes:
    return tag(field, 5) + struct.pack("<f", v)
