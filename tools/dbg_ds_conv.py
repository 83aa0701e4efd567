"""Probe the ResNet downsample/expand 1x1 shapes across tactics: staged
tiles 1-4 vs the direct smallk path, GB/s accounting."""
import os, sys, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch, trtlab_amd
C = trtlab_amd.native()
torch.manual_seed(0)

shapes = [
    ("st2 ds 64->256  M=25088", 8, 56, 56, 64, 256, 1),
    ("st2 c3 64->256  M=25088", 8, 56, 56, 64, 256, 1),
    ("st3 ds 256->512 s2", 8, 56, 56, 256, 512, 2),
    ("st4 ds 512->1024 s2", 8, 28, 28, 512, 1024, 2),
    ("st2 c1 256->64  M=25088", 8, 56, 56, 256, 64, 1),
]
zero = torch.zeros(64, dtype=torch.half, device="cuda")
for name, nb, h, w, cin, cout, s in shapes:
    oh = (h - 1) // s + 1
    M = nb * oh * oh
    x = (torch.randn(nb, h, w, cin, device="cuda") * 0.3).half()
    kp = max(64, ((cin + 63) // 64) * 64)
    wt = (torch.randn(cout, kp, device="cuda") * 0.05).half()
    out = torch.empty(nb, oh, oh, cout, device="cuda").half()
    gb = (M * cin * 2 + cout * cin * 2 + M * cout * 2) / 1e9
    best = []
    for tile in (0, 1, 2, 3, 4):
        torch.cuda.synchronize()
        for _ in range(30):
            C.ops.conv2d(0, x.data_ptr(), wt.data_ptr(), out.data_ptr(),
                         zero_page=zero.data_ptr(), Nb=nb, H=h, W=w, C=cin,
                         Cout=cout, KH=1, KW=1, sh=s, sw=s, ph=0, pw=0,
                         epi=0, tile=tile, sync=False)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        R = 100
        for _ in range(R):
            C.ops.conv2d(0, x.data_ptr(), wt.data_ptr(), out.data_ptr(),
                         zero_page=zero.data_ptr(), Nb=nb, H=h, W=w, C=cin,
                         Cout=cout, KH=1, KW=1, sh=s, sw=s, ph=0, pw=0,
                         epi=0, tile=tile, sync=False)
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / R * 1e6
        best.append((us, tile))
    best.sort()
    row = " ".join(f"t{t}={u:6.1f}" for u, t in sorted(best, key=lambda p: p[1]))
    print(f"{name:26s} {row}  best t{best[0][1]} {best[0][0]:.1f} us "
          f"({gb/best[0][0]*1e6:.0f} GB/s)", flush=True)
