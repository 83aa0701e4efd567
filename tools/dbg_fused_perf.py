import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, torch, trtlab_amd
C = trtlab_amd.native()
torch.manual_seed(0)
M, K = 8, 768
heads, smax = 12, 1024
for N, name in ((2304, "qkv"), (3072, "ff1"), (50257, "head")):
    x = (torch.randn(M, K, device="cuda") * 0.3).half()
    r = torch.randn_like(x).half()
    h_out = torch.zeros_like(x)
    gamma = (torch.rand(K, device="cuda") + 0.5).float()
    beta = torch.randn(K, device="cuda").float() * 0.1
    W = (torch.randn(N, K, device="cuda") * 0.3).half()
    bias = torch.randn(N, device="cuda").float()
    out = torch.zeros(M, N, device="cuda").half()
    kc = torch.zeros(M, heads, smax, 64, device="cuda").half()
    vc = torch.zeros_like(kc)
    pos = torch.full((M,), 5, dtype=torch.int32, device="cuda")
    torch.cuda.synchronize()
    REP = 500
    # warm the clocks so the first-measured case is not penalized
    for _ in range(200):
        C.ops.gemm_bt(0, x.data_ptr(), W.data_ptr(), out.data_ptr(),
                      bias=bias.data_ptr(), M=M, N=N, K=K, epi=1, sync=False)
    torch.cuda.synchronize()
    # unfused gemm_bt (heuristic tiles + scratch)
    t0 = time.perf_counter()
    for _ in range(REP):
        C.ops.gemm_bt(0, x.data_ptr(), W.data_ptr(), out.data_ptr(),
                      bias=bias.data_ptr(), M=M, N=N, K=K, epi=1, sync=False)
    torch.cuda.synchronize()
    t_g = (time.perf_counter() - t0) / REP * 1e6
    # fused
    epi = 2 if name == "qkv" else (1 if name == "ff1" else 0)
    t0 = time.perf_counter()
    for _ in range(REP):
        C.ops.decode_gemm_fused(2, epi, x=x.data_ptr(), r=r.data_ptr(),
                                h_out=h_out.data_ptr(), gamma=gamma.data_ptr(),
                                beta=beta.data_ptr(), B=W.data_ptr(),
                                bias=bias.data_ptr(), C=out.data_ptr(),
                                pos=pos.data_ptr(), kcache=kc.data_ptr(),
                                vcache=vc.data_ptr(), M=M, N=N, K=K,
                                heads=heads, smax=smax, sync=False)
    torch.cuda.synchronize()
    t_f = (time.perf_counter() - t0) / REP * 1e6
    # the ops the fusion replaces: add_layernorm + (kv_append)
    t0 = time.perf_counter()
    for _ in range(REP):
        C.ops.add_layernorm(0, x.data_ptr(), r.data_ptr(), gamma.data_ptr(),
                            beta.data_ptr(), h_out.data_ptr(),
                            sum_out=h_out.data_ptr(), M=M, N=K, sync=False)
    torch.cuda.synchronize()
    t_ln = (time.perf_counter() - t0) / REP * 1e6
    print(f"{name:5s} N={N:6d}: gemm_bt {t_g:7.1f} us  fused {t_f:7.1f} us  "
          f"add_ln {t_ln:6.1f} us", flush=True)
