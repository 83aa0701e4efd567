#!/usr/bin/env python3
"""Pin down the fp4 (cbsz=4) operand layout of the scaled MFMA."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch
import trtlab_amd

C = trtlab_amd.native()

def run(A, B, Sa, Sb):
    a = torch.from_numpy(A).cuda(); b = torch.from_numpy(B).cuda()
    sa = torch.from_numpy(Sa).cuda(); sb = torch.from_numpy(Sb).cuda()
    d = torch.zeros(16, 16, dtype=torch.float32, device="cuda")
    C.ops.mx4_probe(a.data_ptr(), b.data_ptr(), sa.data_ptr(), sb.data_ptr(), d.data_ptr())
    return d.cpu().numpy()

ONE = 0x2   # e2m1 code for 1.0
Z = np.zeros((16, 64), np.uint8)
S1 = np.full((16, 4), 127, np.uint8)

# P0: A elem0 (byte0 low nibble) = 1.0, B elem0 = 1.0 -> 1.0 if low nibble = even elem
A = Z.copy(); A[0, 0] = ONE
B = Z.copy(); B[0, 0] = ONE
print("P0 (low-nibble ~ elem0):", run(A, B, S1, S1)[0, 0], "(expect 1.0)")

# P1: A elem1 (byte0 HIGH nibble) vs B elem1 -> 1.0 if high nibble = odd elem
A = Z.copy(); A[0, 0] = ONE << 4
B = Z.copy(); B[0, 0] = ONE << 4
print("P1 (high-nibble ~ elem1):", run(A, B, S1, S1)[0, 0], "(expect 1.0)")

# P2: cross: A elem0 vs B elem1 -> 0.0 (different k)
A = Z.copy(); A[0, 0] = ONE
B = Z.copy(); B[0, 0] = ONE << 4
print("P2 (cross, expect 0.0):", run(A, B, S1, S1)[0, 0])

# P3: scale block mapping: A k=[0,32) ones, B all ones, Sa[0][0]=128
A = Z.copy(); A[0, :16] = ONE | (ONE << 4)   # elems 0..31
B = Z.copy(); B[0, :] = ONE | (ONE << 4)
Sa = S1.copy(); Sa[0, 0] = 128
print("P3 (expect 64 = 32 elems x2):", run(A, B, Sa, S1)[0, 0])

# P4: full-row sanity: all ones, flat scales -> 128
A = Z.copy(); A[0, :] = ONE | (ONE << 4)
print("P4 (expect 128):", run(A, B, S1, S1)[0, 0])
