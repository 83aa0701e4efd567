import sys, os
sys.path.insert(0, "/root/repo")
import numpy as np, torch, trtlab_amd
from trtlab_amd.engine.mx import quantize_mxfp8, dequantize_mxfp8
C = trtlab_amd.native()

def run(aq, bq, asc, bsc, M, N, K):
    a = torch.from_numpy(aq).cuda(); b = torch.from_numpy(bq).cuda()
    sa = torch.from_numpy(asc).cuda(); sb = torch.from_numpy(bsc).cuda()
    out = torch.empty(M, N, dtype=torch.float32, device="cuda")
    C.ops.gemm_mxfp8(a.data_ptr(), b.data_ptr(), sa.data_ptr(), sb.data_ptr(),
                     out.data_ptr(), M, N, K)
    return out.cpu().numpy()

# frag dump: LDS round-trip check
K0 = 128
A0 = np.arange(64 * K0, dtype=np.int64).astype(np.uint8).reshape(64, K0)
a0 = torch.from_numpy(A0).cuda()
fd = torch.zeros(4 * 64 * 32, dtype=torch.uint8, device="cuda")
C.ops.mx_frag_dump(a0.data_ptr(), fd.data_ptr(), K0)
fd = fd.cpu().numpy().reshape(4, 64, 32)
bad = 0
for wave in range(4):
    for lane in range(64):
        row = wave * 16 + (lane & 15)
        g = lane >> 4
        exp = np.concatenate([A0[row, g*16:(g+1)*16], A0[row, 64+g*16:64+(g+1)*16]])
        if not np.array_equal(fd[wave, lane], exp):
            bad += 1
            if bad < 3:
                print("frag mismatch wave", wave, "lane", lane, fd[wave, lane][:8], exp[:8])
print("frag dump mismatching lanes:", bad, "/ 256")

# single-block full kernel (M=N=64, grid=1)
M = N = 64; K = 128
rng1 = np.random.RandomState(1)
c32 = rng1.randn(M, K).astype(np.float32)
d32 = rng1.randn(N, K).astype(np.float32)
cq, csc = quantize_mxfp8(c32); dq, dsc = quantize_mxfp8(d32)
out1 = run(cq, dq, csc, dsc, M, N, K)
ref1 = dequantize_mxfp8(cq, csc) @ dequantize_mxfp8(dq, dsc).T
print("single block rel err", np.abs(out1 - ref1).max() / np.abs(ref1).max())

M = N = K = 128
rng = np.random.RandomState(0)
a32 = rng.randn(M, K).astype(np.float32)
b32 = rng.randn(N, K).astype(np.float32)

# case 1: flat scales (127): pure fp8 gemm
aq, _ = quantize_mxfp8(a32); bq, _ = quantize_mxfp8(b32)
s127 = np.full((M, K // 32), 127, np.uint8)
out = run(aq, bq, s127, s127, M, N, K)
ref = dequantize_mxfp8(aq, s127) @ dequantize_mxfp8(bq, s127).T
err = np.abs(out - ref).max()
print("flat scales: max err", err, "ref scale", np.abs(ref).max())

# case 2: real block scales
aq, asc = quantize_mxfp8(a32 * np.exp(rng.randn(M, 1)))
bq, bsc = quantize_mxfp8(b32 * np.exp(rng.randn(N, 1)))
out = run(aq, bq, asc, bsc, M, N, K)
ref = dequantize_mxfp8(aq, asc) @ dequantize_mxfp8(bq, bsc).T
err = np.abs(out - ref)
rel = err.max() / np.abs(ref).max()
print("block scales: rel err", rel)
ij = np.unravel_index(err.argmax(), err.shape)
print("worst at", ij, "out", out[ij], "ref", ref[ij])
# which k-tile half mismatches? compute ref per 64-col halves of K
