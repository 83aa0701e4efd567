"""Kernel stress harness: repeated launches with per-launch verdicts to
separate (a) kernel races, (b) upstream corruption, (c) flaky hardware.

  python tools/stress_kernels.py gemm      # 64x64-tile gemm, 30 launches
  python tools/stress_kernels.py att       # attention on BERT-real qkv
  python tools/stress_kernels.py all
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import trtlab_amd

C = trtlab_amd.native()


def verdict(got, want, tag, it):
    got = got.float().cpu()
    err = (got - want).abs()
    nan = int(torch.isnan(got).sum())
    tol = 0.02 + 0.02 * want.abs()
    bad = int(((err > tol) & ~torch.isnan(got)).sum())
    status = "OK" if (nan == 0 and bad <= want.numel() * 1e-4) else "BAD"
    print(f"  {tag} run{it:02d}: {status} nan={nan} bad={bad} "
          f"maxerr={float(err.nan_to_num().max()):.4f} "
          f"last_err={C.hip.last_error()}")
    return status == "OK"


def stress_gemm(iters=30):
    print("== gemm stress (forces 64x64 tiles: M=256,N=512,K=768) ==")
    torch.manual_seed(0)
    M, N, K = 256, 512, 768
    a = torch.randn(M, K, device="cuda").half()
    b = (torch.randn(N, K, device="cuda") * 0.03).half()
    want = a.float() @ b.float().t()
    want = want.cpu()
    ok = 0
    for it in range(iters):
        out = torch.full((M, N), float("nan"), dtype=torch.half, device="cuda")
        torch.cuda.synchronize()
        C.ops.gemm_bt(0, a.data_ptr(), b.data_ptr(), out.data_ptr(),
                      M=M, N=N, K=K, epi=0)
        ok += verdict(out, want, "gemm", it)
    print(f"gemm: {ok}/{iters} ok")


def stress_att(iters=30):
    print("== attention stress (CPU-computed real qkv) ==")
    from trtlab_amd.engine.planner import Planner
    from trtlab_amd.engine.reference import run_reference
    from trtlab_amd.models import build_bert

    B, S, H, D = 2, 128, 12, 64
    hid = H * D
    plan = Planner(reuse=False).compile(build_bert(batch=B, seq=S, layers=1, seed=0))
    x = np.random.RandomState(9).randn(*plan.input_shape).astype(np.float32)
    cpu = run_reference(plan, x, return_all=True)
    qkv16 = cpu["l0_qkv"].astype(np.float16)
    print("qkv nan:", int(np.isnan(qkv16).sum()))
    qkv_t = torch.from_numpy(qkv16).cuda().contiguous()
    q = torch.from_numpy(qkv16.astype(np.float32)).reshape(B, S, 3, H, D)
    qq, kk, vv = (q[:, :, i].permute(0, 2, 1, 3) for i in range(3))
    att = torch.softmax(qq @ kk.transpose(-1, -2) / np.sqrt(D), dim=-1)
    want = (att @ vv).permute(0, 2, 1, 3).reshape(B * S, hid)
    ok = 0
    for it in range(iters):
        out = torch.full((B * S, hid), float("nan"), dtype=torch.half,
                         device="cuda")
        torch.cuda.synchronize()
        C.ops.attention(0, qkv_t.data_ptr(), out.data_ptr(), B, S, H, D,
                        float(1.0 / np.sqrt(D)))
        ok += verdict(out, want, "att", it)
    print(f"att: {ok}/{iters} ok")


def stress_conv(iters=20):
    print("== conv stress (64x64 tiles: stage5-like 392x512 K=4608) ==")
    torch.manual_seed(1)
    nb, h, w, cin, cout = 8, 7, 7, 512, 512
    x = torch.randn(nb, h, w, cin, device="cuda").half()
    wt = (torch.randn(cout, cin, 3, 3, device="cuda") * 0.02).half()
    flat = wt.permute(0, 2, 3, 1).reshape(cout, 9 * cin)
    kp = ((9 * cin + 63) // 64) * 64
    if kp != flat.shape[1]:
        flat = torch.nn.functional.pad(flat, (0, kp - flat.shape[1]))
    wp = flat.half().contiguous()
    zero = torch.zeros(64, dtype=torch.half, device="cuda")
    want = torch.nn.functional.conv2d(
        x.float().permute(0, 3, 1, 2), wt.float(), padding=1
    ).permute(0, 2, 3, 1).cpu()
    ok = 0
    for it in range(iters):
        out = torch.full((nb, h, w, cout), float("nan"), dtype=torch.half,
                         device="cuda")
        torch.cuda.synchronize()
        C.ops.conv2d(0, x.data_ptr(), wp.data_ptr(), out.data_ptr(),
                     zero_page=zero.data_ptr(), Nb=nb, H=h, W=w, C=cin,
                     Cout=cout, KH=3, KW=3, sh=1, sw=1, ph=1, pw=1, epi=0)
        ok += verdict(out, want, "conv", it)
    print(f"conv: {ok}/{iters} ok")


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    if which in ("gemm", "all"):
        stress_gemm()
    if which in ("conv", "all"):
        stress_conv()
    if which in ("att", "all"):
        stress_att()
