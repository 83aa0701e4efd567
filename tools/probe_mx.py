#!/usr/bin/env python3
"""Empirically pin down mfma_scale_f32_16x16x128_f8f6f4 operand layout."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import trtlab_amd

C = trtlab_amd.native()
ONE = np.float32(1.0)

def fp8(x):
    return torch.tensor(x, dtype=torch.float32).to(torch.float8_e4m3fn).view(torch.uint8).item()

def run(A, B, Sa, Sb):
    a = torch.from_numpy(A).cuda(); b = torch.from_numpy(B).cuda()
    sa = torch.from_numpy(Sa).cuda(); sb = torch.from_numpy(Sb).cuda()
    d = torch.zeros(16, 16, dtype=torch.float32, device="cuda")
    C.ops.mx_probe(a.data_ptr(), b.data_ptr(), sa.data_ptr(), sb.data_ptr(), d.data_ptr())
    return d.cpu().numpy()

one = fp8(1.0); two = fp8(2.0)
Z = np.zeros((16, 128), np.uint8)
S1 = np.full((16, 4), 127, np.uint8)

# E0: single element A[0][0]=1, B[0][0]=1 -> D[0][0] should be 1.0 (fmt check)
A = Z.copy(); A[0, 0] = one
B = Z.copy(); B[0, 0] = one
d = run(A, B, S1, S1)
print("E0 D[0,0] =", d[0, 0], " (expect 1.0 if cbsz=0 is e4m3)")

# E1: A row0 = 1 for k in [0,32), B row0 all 1; Sa[0][0]=128 (2.0)
A = Z.copy(); A[0, :32] = one
B = Z.copy(); B[0, :] = one
Sa = S1.copy(); Sa[0, 0] = 128
d = run(A, B, Sa, S1)
print("E1 D[0,0] =", d[0, 0], " (64 => lane g covers k=[g*32,+32) & scale=Sa[row][g];"
      " 48 => split-half layout)")

# E2: which k positions does lane-g's 32 bytes cover? bump one byte at a time
for kx in (0, 15, 16, 31, 32, 63, 64, 96, 127):
    A = Z.copy(); A[0, kx] = two
    B = Z.copy(); B[0, :] = one
    d = run(A, B, S1, S1)
    print(f"E2 kx={kx:3d} D[0,0]={d[0,0]:.1f} (2.0 everywhere => all k reach row sum)")

# E3: diag check - B rows map to cols?
A = Z.copy(); A[:, 0] = one
B = Z.copy(); B[3, 0] = one
d = run(A, B, S1, S1)
print("E3 nonzero cols for all rows:", np.nonzero(d[0])[0], "(expect col 3)")
