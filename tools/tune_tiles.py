"""Tile-config sweep over the actual ResNet-50 b8 + BERT-base b8 shapes.
For each (conv/gemm) shape, time all four BMxBN configs (50 reps each,
stream-synced batches) and print the winner vs the pick_tile heuristic.
Run on the GPU box; the result drives the pick_tile scoring.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import trtlab_amd

C = trtlab_amd.native()

TILES = {1: "128x128", 2: "128x64", 3: "64x128", 4: "64x64"}


def time_launch(fn, reps=50, warmup=10):
    for _ in range(warmup):
        fn(sync=False)
    torch.cuda.synchronize()
    C.hip.device_synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn(sync=False)
    C.hip.device_synchronize()
    return (time.perf_counter() - t0) / reps * 1e6  # us


def sweep_conv(name, nb, h, w, cin, cout, k, s, p):
    x = torch.randn(nb, h, w, cin, device="cuda").half()
    flat = (torch.randn(cout, k * k * cin, device="cuda") * 0.02)
    kp = ((k * k * cin + 63) // 64) * 64
    if kp != flat.shape[1]:
        flat = torch.nn.functional.pad(flat, (0, kp - flat.shape[1]))
    wp = flat.half().contiguous()
    zero = torch.zeros(64, dtype=torch.half, device="cuda")
    oh = (h + 2 * p - k) // s + 1
    ow = (w + 2 * p - k) // s + 1
    out = torch.empty(nb, oh, ow, cout, dtype=torch.half, device="cuda")
    res = []
    for tile in (1, 2, 3, 4):
        us = time_launch(lambda sync: C.ops.conv2d(
            0, x.data_ptr(), wp.data_ptr(), out.data_ptr(),
            zero_page=zero.data_ptr(), Nb=nb, H=h, W=w, C=cin, Cout=cout,
            KH=k, KW=k, sh=s, sw=s, ph=p, pw=p, epi=0, sync=sync, tile=tile))
        res.append((us, tile))
    auto = time_launch(lambda sync: C.ops.conv2d(
        0, x.data_ptr(), wp.data_ptr(), out.data_ptr(),
        zero_page=zero.data_ptr(), Nb=nb, H=h, W=w, C=cin, Cout=cout,
        KH=k, KW=k, sh=s, sw=s, ph=p, pw=p, epi=0, sync=sync, tile=0))
    res.sort()
    flops = 2.0 * nb * oh * ow * cout * k * k * cin
    best_us, best_t = res[0]
    print(f"conv {name:26} M={nb*oh*ow:6d} N={cout:4d} K={k*k*cin:5d} | " +
          " ".join(f"{TILES[t]}={u:7.1f}us" for u, t in sorted(res, key=lambda r: r[1])) +
          f" | best={TILES[best_t]} {flops/best_us/1e6:6.1f}TF auto={auto:7.1f}us")


def sweep_gemm(name, M, N, K):
    a = torch.randn(M, K, device="cuda").half()
    b = (torch.randn(N, K, device="cuda") * 0.03).half()
    out = torch.empty(M, N, dtype=torch.half, device="cuda")
    res = []
    for tile in (1, 2, 3, 4):
        us = time_launch(lambda sync: C.ops.gemm_bt(
            0, a.data_ptr(), b.data_ptr(), out.data_ptr(), M=M, N=N, K=K,
            epi=0, sync=sync, tile=tile))
        res.append((us, tile))
    auto = time_launch(lambda sync: C.ops.gemm_bt(
        0, a.data_ptr(), b.data_ptr(), out.data_ptr(), M=M, N=N, K=K, epi=0,
        sync=sync, tile=0))
    res.sort()
    flops = 2.0 * M * N * K
    best_us, best_t = res[0]
    print(f"gemm {name:26} M={M:6d} N={N:4d} K={K:5d} | " +
          " ".join(f"{TILES[t]}={u:7.1f}us" for u, t in sorted(res, key=lambda r: r[1])) +
          f" | best={TILES[best_t]} {flops/best_us/1e6:6.1f}TF auto={auto:7.1f}us")


B = 8
print("=== ResNet-50 b8 conv shapes ===")
sweep_conv("stem 7x7s2", B, 224, 224, 8, 64, 7, 2, 3)
sweep_conv("s1 1x1 64->64", B, 56, 56, 64, 64, 1, 1, 0)
sweep_conv("s1 3x3 64", B, 56, 56, 64, 64, 3, 1, 1)
sweep_conv("s1 1x1 64->256", B, 56, 56, 64, 256, 1, 1, 0)
sweep_conv("s1 1x1 256->64", B, 56, 56, 256, 64, 1, 1, 0)
sweep_conv("s2 1x1 256->128", B, 56, 56, 256, 128, 1, 1, 0)
sweep_conv("s2 3x3 128 s2", B, 56, 56, 128, 128, 3, 2, 1)
sweep_conv("s2 3x3 128", B, 28, 28, 128, 128, 3, 1, 1)
sweep_conv("s2 1x1 128->512", B, 28, 28, 128, 512, 1, 1, 0)
sweep_conv("s2 1x1 512->128", B, 28, 28, 512, 128, 1, 1, 0)
sweep_conv("s3 3x3 256", B, 14, 14, 256, 256, 3, 1, 1)
sweep_conv("s3 1x1 256->1024", B, 14, 14, 256, 1024, 1, 1, 0)
sweep_conv("s3 1x1 1024->256", B, 14, 14, 1024, 256, 1, 1, 0)
sweep_conv("s4 3x3 512", B, 7, 7, 512, 512, 3, 1, 1)
sweep_conv("s4 1x1 512->2048", B, 7, 7, 512, 2048, 1, 1, 0)
sweep_conv("s4 1x1 2048->512", B, 7, 7, 2048, 512, 1, 1, 0)
print("=== BERT-base b8 gemm shapes (M=1024) ===")
sweep_gemm("qkv", 1024, 2304, 768)
sweep_gemm("proj", 1024, 768, 768)
sweep_gemm("ff1", 1024, 3072, 768)
sweep_gemm("ff2", 1024, 768, 3072)
print("=== square reference ===")
sweep_gemm("4096^3", 4096, 4096, 4096)
