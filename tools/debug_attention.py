"""Attention kernel deep-dive: compare the engine's attention output, a
standalone kernel run on the same qkv data, and a torch fp32 reference.
Prints error structure (bad fraction, distribution over rows/cols/waves)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

import trtlab_amd
from trtlab_amd.engine.planner import Planner
from trtlab_amd.engine.reference import run_reference
from trtlab_amd.engine.runtime import NativeEngine
from trtlab_amd.models import build_bert

C = trtlab_amd.native()

B, S, H, D = 2, 128, 12, 64
hid = H * D

plan = Planner(reuse=False).compile(build_bert(batch=B, seq=S, layers=2, seed=0))
x = np.random.RandomState(9).randn(*plan.input_shape).astype(np.float32)
cpu = run_reference(plan, x, return_all=True)

eng = NativeEngine(plan)
ctx = eng.create_context(capture=False)
ctx.infer(x)
arena = ctx.ctx.arena_ptr

def read(t):
    shape = plan.shapes[t]
    buf = np.empty(int(np.prod(shape)), dtype=np.float16)
    C.memory.memcpy_d2h(buf, arena + plan.offsets[t], buf.nbytes)
    return buf.reshape(shape).astype(np.float32)

qkv_gpu = read("l0_qkv")        # engine's actual fp16 qkv
att_eng = read("l0_att")        # engine's attention output

# torch fp32 reference on the SAME fp16 qkv
q = torch.from_numpy(qkv_gpu).reshape(B, S, 3, H, D)
qq = q[:, :, 0].permute(0, 2, 1, 3)
kk = q[:, :, 1].permute(0, 2, 1, 3)
vv = q[:, :, 2].permute(0, 2, 1, 3)
att = torch.softmax(qq @ kk.transpose(-1, -2) / np.sqrt(D), dim=-1)
ref = (att @ vv).permute(0, 2, 1, 3).reshape(B * S, hid).numpy()

# standalone kernel on the same qkv bytes
qkv_t = torch.from_numpy(qkv_gpu).half().cuda().contiguous()
out_t = torch.empty(B * S, hid, dtype=torch.half, device="cuda")
torch.cuda.synchronize()
C.ops.attention(0, qkv_t.data_ptr(), out_t.data_ptr(), B, S, H, D,
                float(1.0 / np.sqrt(D)))
att_alone = out_t.float().cpu().numpy()


def report(name, got, want):
    err = np.abs(got - want)
    scale = np.abs(want).max()
    tol = 0.02 + 0.02 * np.abs(want)
    bad = err > tol
    print(f"{name}: max={err.max():.4f} rel={err.max()/scale:.4f} "
          f"badfrac={bad.mean():.5f} nbad={bad.sum()}")
    if bad.sum():
        rows, cols = np.where(bad)
        print(f"  bad rows: min={rows.min()} max={rows.max()} "
              f"uniq_mod32={sorted(set((rows % 32).tolist()))[:12]}")
        print(f"  bad cols: heads={sorted(set((cols // D).tolist()))} "
              f"d_mod16={sorted(set((cols % 16).tolist()))[:18]}")
        print(f"  rows mod 4: {sorted(set((rows % 4).tolist()))} "
              f" b: {sorted(set((rows // S).tolist()))}")


report("engine att  vs torch(fp16qkv)", att_eng, ref)
report("standalone  vs torch(fp16qkv)", att_alone, ref)
report("engine      vs standalone    ", att_eng, att_alone)
report("cpu_ref att vs torch(fp16qkv)", cpu["l0_att"], ref)

# determinism check: run the standalone kernel 10x, compare runs
print("\ndeterminism check (10 runs):")
outs = []
for it in range(10):
    o = torch.empty(B * S, hid, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.attention(0, qkv_t.data_ptr(), o.data_ptr(), B, S, H, D,
                    float(1.0 / np.sqrt(D)))
    outs.append(o.float().cpu().numpy())
ref0 = outs[0]
for it, o in enumerate(outs[1:], 1):
    same = np.array_equal(o, ref0, equal_nan=True)
    nans = int(np.isnan(o).sum())
    err = np.nanmax(np.abs(o - ref))
    print(f"  run{it}: identical_to_run0={same} nans={nans} max_vs_torch={err:.4f}")
