import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch, trtlab_amd
from trtlab_amd.engine.mx import quantize_mxfp4, dequantize_mxfp4
C = trtlab_amd.native()

def run(aq, bq, asc, bsc, M, N, K):
    a = torch.from_numpy(aq).cuda(); b = torch.from_numpy(bq).cuda()
    sa = torch.from_numpy(asc).cuda(); sb = torch.from_numpy(bsc).cuda()
    out = torch.empty(M, N, dtype=torch.float32, device="cuda")
    C.ops.gemm_mxfp4(a.data_ptr(), b.data_ptr(), sa.data_ptr(), sb.data_ptr(),
                     out.data_ptr(), M, N, K)
    return out.cpu().numpy()

M = N = 128; K = 256
rng = np.random.RandomState(0)
a32 = rng.randn(M, K).astype(np.float32)
b32 = rng.randn(N, K).astype(np.float32)
aq, asc = quantize_mxfp4(a32); bq, bsc = quantize_mxfp4(b32)
out = run(aq, bq, asc, bsc, M, N, K)
ref = dequantize_mxfp4(aq, asc) @ dequantize_mxfp4(bq, bsc).T
err = np.abs(out - ref)
print("max err", err.max(), "ref scale", np.abs(ref).max(),
      "frac bad", (err > 0.01 + 0.01*np.abs(ref)).mean())
ij = np.unravel_index(err.argmax(), err.shape)
print("worst", ij, out[ij], ref[ij])
# localize: zero second window (k 128:256) -> single-window check
aq2 = aq.copy(); aq2[:, 64:] = 0
bq2 = bq.copy(); bq2[:, 64:] = 0
out = run(aq2, bq2, asc, bsc, M, N, K)
ref = dequantize_mxfp4(aq2, asc) @ dequantize_mxfp4(bq2, bsc).T
print("window0 only: max err", np.abs(out - ref).max())
# zero first window
aq3 = aq.copy(); aq3[:, :64] = 0
bq3 = bq.copy(); bq3[:, :64] = 0
out = run(aq3, bq3, asc, bsc, M, N, K)
ref = dequantize_mxfp4(aq3, asc) @ dequantize_mxfp4(bq3, bsc).T
print("window1 only: max err", np.abs(out - ref).max())
