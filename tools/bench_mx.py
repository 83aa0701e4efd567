#!/usr/bin/env python3
"""MXFP8 GEMM micro-benchmark (CDNA4 scaled MFMA 16x16x128).

Times square GEMMs and reports effective TFLOP/s next to the fp16
16x16x32 path for context. Run on an MI355X:
    python tools/bench_mx.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import trtlab_amd
from trtlab_amd.engine.mx import quantize_mxfp4, quantize_mxfp8

C = trtlab_amd.native()


def time_op(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    for size in (2048, 4096, 8192):
        M = N = K = size
        rng = np.random.RandomState(0)
        a32 = rng.randn(M, K).astype(np.float32) * 0.5
        b32 = rng.randn(N, K).astype(np.float32) * 0.5
        aq, asc = quantize_mxfp8(a32)
        bq, bsc = quantize_mxfp8(b32)
        a = torch.from_numpy(aq).cuda()
        b = torch.from_numpy(bq).cuda()
        sa = torch.from_numpy(asc).cuda()
        sb = torch.from_numpy(bsc).cuda()
        out = torch.empty(M, N, dtype=torch.float32, device="cuda")

        tmx = time_op(lambda: C.ops.gemm_mxfp8(
            a.data_ptr(), b.data_ptr(), sa.data_ptr(), sb.data_ptr(),
            out.data_ptr(), M, N, K, 0, False))

        a4, as4 = quantize_mxfp4(a32)
        b4, bs4 = quantize_mxfp4(b32)
        a4t = torch.from_numpy(a4).cuda()
        b4t = torch.from_numpy(b4).cuda()
        sa4 = torch.from_numpy(as4).cuda()
        sb4 = torch.from_numpy(bs4).cuda()
        tmx4 = time_op(lambda: C.ops.gemm_mxfp4(
            a4t.data_ptr(), b4t.data_ptr(), sa4.data_ptr(), sb4.data_ptr(),
            out.data_ptr(), M, N, K, 0, False))

        ah = torch.from_numpy(a32).half().cuda()
        bh = torch.from_numpy(b32).half().cuda()
        oh = torch.empty(M, N, dtype=torch.half, device="cuda")
        tfp16 = time_op(lambda: C.ops.gemm_bt(
            0, ah.data_ptr(), bh.data_ptr(), oh.data_ptr(),
            M=M, N=N, K=K, sync=False))

        flops = 2.0 * M * N * K
        print(f"{size}^3: mxfp8 {tmx*1e3:7.3f} ms = {flops/tmx/1e12:7.1f} TF"
              f" | mxfp4 {tmx4*1e3:7.3f} ms = {flops/tmx4/1e12:7.1f} TF"
              f" | fp16 {tfp16*1e3:7.3f} ms = {flops/tfp16/1e12:7.1f} TF",
              flush=True)


if __name__ == "__main__":
    main()
