"""Fused bottleneck-tail kernel: numerics vs the 2-kernel composition +
microbenchmark at the ResNet-50 b8 stage-1/2 shapes.

Run on the GPU box:  python tools/dbg_btail.py
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import trtlab_amd

C = trtlab_amd.native()


def prep_w1(w):  # [Cm, Cm, 3, 3] OIHW -> bt [Cm][9*Cm] (kh, kw, ci) order
    co, ci, kh, kw = w.shape
    return np.ascontiguousarray(
        w.transpose(0, 2, 3, 1).reshape(co, kh * kw * ci))


def run_case(nb, h, w, cm, co, iters=200):
    rng = np.random.RandomState(0)
    x = torch.from_numpy(rng.randn(nb, h, w, cm).astype(np.float32) * 0.5) \
        .half().cuda()
    w1 = rng.randn(cm, cm, 3, 3).astype(np.float32) * (1.0 / np.sqrt(9 * cm))
    w2 = rng.randn(co, cm).astype(np.float32) * (1.0 / np.sqrt(cm))
    s1 = rng.uniform(0.5, 1.5, cm).astype(np.float32)
    b1 = rng.randn(cm).astype(np.float32) * 0.1
    s2 = rng.uniform(0.5, 1.5, co).astype(np.float32)
    b2 = rng.randn(co).astype(np.float32) * 0.1
    res = torch.from_numpy(rng.randn(nb, h, w, co).astype(np.float32) * 0.5) \
        .half().cuda()

    dw1 = torch.from_numpy(prep_w1(w1)).half().cuda()
    dw2 = torch.from_numpy(w2).half().cuda()
    ds1, db1 = (torch.from_numpy(s1).cuda(), torch.from_numpy(b1).cuda())
    ds2, db2 = (torch.from_numpy(s2).cuda(), torch.from_numpy(b2).cuda())
    zero = torch.zeros(64, dtype=torch.half, device="cuda")
    t_mid = torch.empty(nb, h, w, cm, dtype=torch.half, device="cuda")
    out_ref = torch.empty(nb, h, w, co, dtype=torch.half, device="cuda")
    out_fused = torch.empty_like(out_ref)

    def two_kernel(sync=True):
        C.ops.conv2d(0, x.data_ptr(), dw1.data_ptr(), t_mid.data_ptr(),
                     scale=ds1.data_ptr(), bias=db1.data_ptr(),
                     zero_page=zero.data_ptr(), Nb=nb, H=h, W=w, C=cm,
                     Cout=cm, KH=3, KW=3, sh=1, sw=1, ph=1, pw=1, epi=5,
                     sync=False)
        C.ops.conv2d(0, t_mid.data_ptr(), dw2.data_ptr(),
                     out_ref.data_ptr(), scale=ds2.data_ptr(),
                     bias=db2.data_ptr(), residual=res.data_ptr(),
                     zero_page=zero.data_ptr(), Nb=nb, H=h, W=w, C=cm,
                     Cout=co, KH=1, KW=1, epi=6, sync=sync)

    def fused(sync=True):
        C.ops.bottleneck_tail(0, x.data_ptr(), dw1.data_ptr(),
                              dw2.data_ptr(), out_fused.data_ptr(),
                              ds1.data_ptr(), db1.data_ptr(),
                              ds2.data_ptr(), db2.data_ptr(),
                              res.data_ptr(), zero.data_ptr(), Nb=nb, H=h,
                              W=w, Cm=cm, Co=co, sync=sync)

    two_kernel()
    fused()
    a = out_ref.float().cpu().numpy()
    b = out_fused.float().cpu().numpy()
    err = np.abs(a - b).max() / max(np.abs(a).max(), 1e-6)
    print(f"  [{nb}x{h}x{w} Cm={cm} Co={co}] rel err vs 2-kernel: {err:.5f}")
    assert err < 0.02, err

    def tim(fn):
        for _ in range(20):
            fn(sync=False)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(iters):
            fn(sync=False)
        torch.cuda.synchronize()
        return (time.perf_counter() - t0) / iters * 1e6

    t2 = tim(two_kernel)
    tf = tim(fused)
    print(f"    two-kernel {t2:8.1f} us   fused {tf:8.1f} us   "
          f"{'WIN' if tf < t2 else 'LOSS'} {t2 / tf:.2f}x")
    return t2, tf


if __name__ == "__main__":
    print("bottleneck-tail fusion (b8 rn50 shapes):")
    run_case(8, 56, 56, 64, 256)    # stage 1 (3 pairs / net)
    run_case(8, 28, 28, 128, 512)   # stage 2 (4 pairs / net)
