"""One-shot GPU diagnostics: device info, MFMA fragment-layout probe,
kernel spot checks. Run on the GPU box; prints everything needed to debug a
wrong fragment-layout guess without a second round trip."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import trtlab_amd

C = trtlab_amd.native()
print("devices:", C.hip.device_count())
print(C.hip.device_properties(0))

EPS = 0


def gemm(a, b, M, N, K):
    out = torch.empty(M, N, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.gemm_bt(0, a.data_ptr(), b.data_ptr(), out.data_ptr(), M=M, N=N,
                  K=K, epi=0)
    return out


# ---- probe 1: one-hot A rows x structured B ----
M = N = 16
K = 64
a = torch.zeros(M, K, dtype=torch.half, device="cuda")
# A[i][k] = (i == 2 && k == 5) -> out[2][n] should equal B[n][5]
a[2, 5] = 1.0
b = torch.zeros(N, K, dtype=torch.half, device="cuda")
for n in range(N):
    for k in range(8):
        b[n, k] = n + k * 0.0625  # asymmetric
out = gemm(a, b, M, N, K).float().cpu()
expect = b[:, 5].float().cpu()
print("probe1 row2:", out[2, :8].tolist())
print("probe1 want:", expect[:8].tolist())
print("probe1 other rows max:", out[torch.arange(16) != 2].abs().max().item())

# ---- probe 2: dense random check with error structure ----
torch.manual_seed(0)
M, N, K = 128, 128, 64
a = torch.randn(M, K, device="cuda").half()
b = torch.randn(N, K, device="cuda").half()
out = gemm(a, b, M, N, K).float().cpu()
ref = (a.float() @ b.float().t()).cpu()
err = (out - ref).abs()
print("probe2 max err:", err.max().item(), "mean:", err.mean().item())
if err.max() > 0.5:
    # where is it wrong? print error heatmap coarse 8x8
    h = err.reshape(8, 16, 8, 16).amax(dim=(1, 3))
    print("err heatmap 8x8 (16-blocks):")
    print(np.array2string(h.numpy(), precision=1))
    # is it a transpose?
    errT = (out - ref.t()).abs()
    print("transpose hypothesis max err:", errT.max().item())

# ---- probe 3: larger K, tails ----
M, N, K = 200, 300, 256
a = torch.randn(M, K, device="cuda").half()
b = torch.randn(N, K, device="cuda").half()
out = gemm(a, b, M, N, K).float().cpu()
ref = (a.float() @ b.float().t()).cpu()
err = (out - ref).abs()
tol = 0.02 + 0.02 * ref.abs()
print("probe3 bad frac:", (err > tol).float().mean().item(),
      "max err:", err.max().item())

# ---- probe 4: conv 1x1 ----
x = torch.randn(2, 8, 8, 64, device="cuda").half()
w = (torch.randn(128, 64, 1, 1, device="cuda") * 0.1).half()
wp = w.permute(0, 2, 3, 1).reshape(128, 64).contiguous()
zero = torch.zeros(64, dtype=torch.half, device="cuda")
out = torch.empty(2, 8, 8, 128, dtype=torch.half, device="cuda")
torch.cuda.synchronize()
C.ops.conv2d(0, x.data_ptr(), wp.data_ptr(), out.data_ptr(),
             zero_page=zero.data_ptr(), Nb=2, H=8, W=8, C=64, Cout=128,
             KH=1, KW=1, sh=1, sw=1, ph=0, pw=0, epi=0)
ref = torch.nn.functional.conv2d(x.float().permute(0, 3, 1, 2), w.float())
ref = ref.permute(0, 2, 3, 1).cpu()
err = (out.float().cpu() - ref).abs()
print("probe4 conv1x1 max err:", err.max().item())

# ---- probe 5: conv 3x3 pad ----
x = torch.randn(2, 8, 8, 64, device="cuda").half()
w = (torch.randn(64, 64, 3, 3, device="cuda") * 0.05).half()
wp = w.permute(0, 2, 3, 1).reshape(64, 9 * 64).contiguous()
out = torch.empty(2, 8, 8, 64, dtype=torch.half, device="cuda")
torch.cuda.synchronize()
C.ops.conv2d(0, x.data_ptr(), wp.data_ptr(), out.data_ptr(),
             zero_page=zero.data_ptr(), Nb=2, H=8, W=8, C=64, Cout=64,
             KH=3, KW=3, sh=1, sw=1, ph=1, pw=1, epi=0)
ref = torch.nn.functional.conv2d(x.float().permute(0, 3, 1, 2), w.float(),
                                 padding=1).permute(0, 2, 3, 1).cpu()
err = (out.float().cpu() - ref).abs()
print("probe5 conv3x3 max err:", err.max().item())
print("ALL PROBES DONE")
