"""OCP Microscaling (MX) format host-side codecs.

MXFP8: fp8-e4m3 elements with one shared e8m0 scale per 32-element block
(along K). The GPU side is csrc/kernels/gemm_mx.hip on the CDNA4 scaled
MFMA (mfma_scale_f32_16x16x128_f8f6f4) — gfx950's highest-throughput GEMM
instruction, with no equivalent in the CUDA reference. Engine-level MX
plans are round-2 work; this module + the raw op are the validated
foundation (tests/test_kernels_gpu.py, tools/bench_mx.py).
"""
from __future__ import annotations

import numpy as np
import torch

BLOCK = 32  # OCP MX block size
_E4M3_EMAX = 8  # e4m3 max binade (448 = 1.75 * 2^8)


def quantize_mxfp8(x: np.ndarray) -> tuple[np.ndarray, np.ndarray]:
    """[R, K] fp32 -> (codes u8 [R, K], scales u8 [R, K/32]).

    Per-block shared scale 2^(floor(log2(amax)) - 8) per the OCP MX spec
    (element format e4m3, emax 8); e8m0 byte = exponent + 127.
    """
    r, k = x.shape
    assert k % BLOCK == 0, "K must be a multiple of the MX block (32)"
    t = torch.from_numpy(np.ascontiguousarray(x, np.float32))
    blocks = t.reshape(r, k // BLOCK, BLOCK)
    amax = blocks.abs().amax(dim=2)
    e = torch.where(amax > 0, torch.floor(torch.log2(amax)),
                    torch.zeros_like(amax)) - _E4M3_EMAX
    e = torch.clamp(e, -127, 127)
    scale = torch.pow(2.0, e)
    q = torch.clamp(blocks / scale[:, :, None], -448, 448)
    codes = q.to(torch.float8_e4m3fn).view(torch.uint8)
    scales = (e + 127).to(torch.uint8)
    return (codes.reshape(r, k).numpy().copy(), scales.numpy().copy())


def dequantize_mxfp8(codes: np.ndarray, scales: np.ndarray) -> np.ndarray:
    """Inverse of quantize_mxfp8 (the CPU oracle for the GPU kernel)."""
    r, k = codes.shape
    vals = torch.from_numpy(codes).view(torch.float8_e4m3fn).float()
    scale = torch.pow(2.0, torch.from_numpy(scales).float() - 127)
    out = vals.reshape(r, k // BLOCK, BLOCK) * scale[:, :, None]
    return out.reshape(r, k).numpy()


# ---- MXFP4 (fp4 e2m1 elements, e8m0 block scales) ----
_E2M1_VALS = [0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0]
_E2M1_EMAX = 2  # 6 = 1.5 * 2^2


def quantize_mxfp4(x: np.ndarray) -> tuple[np.ndarray, np.ndarray]:
    """[R, K] fp32 -> (packed u8 [R, K/2] (low nibble = even element),
    scales u8 [R, K/32])."""
    r, k = x.shape
    assert k % BLOCK == 0
    t = torch.from_numpy(np.ascontiguousarray(x, np.float32))
    blocks = t.reshape(r, k // BLOCK, BLOCK)
    amax = blocks.abs().amax(dim=2)
    e = torch.where(amax > 0, torch.floor(torch.log2(amax)),
                    torch.zeros_like(amax)) - _E2M1_EMAX
    e = torch.clamp(e, -127, 127)
    scale = torch.pow(2.0, e)
    q = blocks / scale[:, :, None]
    vals = torch.tensor(_E2M1_VALS)
    mids = (vals[1:] + vals[:-1]) / 2  # round-to-nearest boundaries
    mag = torch.bucketize(q.abs().reshape(-1), mids).to(torch.uint8)
    code = torch.where(q.reshape(-1) < 0, mag | 8, mag).reshape(r, k)
    packed = (code[:, 0::2] | (code[:, 1::2] << 4)).to(torch.uint8)
    return packed.numpy().copy(), (e + 127).to(torch.uint8).numpy().copy()


def dequantize_mxfp4(packed: np.ndarray, scales: np.ndarray) -> np.ndarray:
    r, kh = packed.shape
    k = kh * 2
    p = torch.from_numpy(packed)
    code = torch.empty(r, k, dtype=torch.uint8)
    code[:, 0::2] = p & 15
    code[:, 1::2] = p >> 4
    vals = torch.tensor(_E2M1_VALS)
    mag = vals[(code & 7).long()]
    v = torch.where(code >= 8, -mag, mag)
    scale = torch.pow(2.0, torch.from_numpy(scales).float() - 127)
    return (v.reshape(r, k // BLOCK, BLOCK) * scale[:, :, None]).reshape(
        r, k).numpy()
