"""GPU kernel numerics: each HIP kernel vs a plain PyTorch fp32 reference.

All tensors fp16 on device; references computed in fp32. Asymmetric random
inputs (transpose-detecting — cdna guide §5.4 rule 16).
"""
import numpy as np
import pytest
import torch

import trtlab_amd

pytestmark = pytest.mark.gpu

# epilogue codes (csrc/kernels/gemm_common.h)
EPI_NONE, EPI_BIAS, EPI_BIAS_RELU, EPI_BIAS_GELU = 0, 1, 2, 3
EPI_SB, EPI_SB_RELU, EPI_SB_ADD_RELU = 4, 5, 6


@pytest.fixture(scope="module")
def C():
    return trtlab_amd.native()


def t16(*shape, seed=0, scale=1.0):
    g = torch.Generator(device="cuda").manual_seed(seed)
    return (torch.randn(*shape, generator=g, device="cuda") * scale).half().contiguous()


def check(out, ref, rtol=2e-2, atol=2e-2, frac=1e-3):
    out = out.float().cpu()
    ref = ref.float().cpu()
    err = (out - ref).abs()
    tol = atol + rtol * ref.abs()
    bad = (err > tol).float().mean().item()
    assert bad <= frac, f"mismatch frac {bad}: max err {err.max()} vs {ref.abs().max()}"


@pytest.mark.parametrize("M,N,K", [(128, 128, 64), (256, 384, 128),
                                   (1000, 1000, 256), (100, 64, 2048),
                                   (8, 1000, 2048)])
def test_gemm_bt(C, M, N, K):
    a = t16(M, K, seed=M + N)
    b = t16(N, K, seed=M + N + 1)
    out = torch.empty(M, N, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.gemm_bt(0, a.data_ptr(), b.data_ptr(), out.data_ptr(),
                  M=M, N=N, K=K, epi=EPI_NONE)
    ref = a.float() @ b.float().t()
    check(out, ref)


def test_gemm_bt_bias_relu(C):
    M, N, K = 200, 300, 192
    a, b = t16(M, K, seed=1), t16(N, K, seed=2)
    bias = torch.randn(N, device="cuda").float().contiguous()
    out = torch.empty(M, N, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.gemm_bt(0, a.data_ptr(), b.data_ptr(), out.data_ptr(),
                  bias=bias.data_ptr(), M=M, N=N, K=K, epi=EPI_BIAS_RELU)
    ref = torch.relu(a.float() @ b.float().t() + bias)
    check(out, ref)


def test_gemm_bt_bias_gelu(C):
    M, N, K = 128, 256, 128
    a, b = t16(M, K, seed=3), t16(N, K, seed=4)
    bias = torch.randn(N, device="cuda").float().contiguous()
    out = torch.empty(M, N, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.gemm_bt(0, a.data_ptr(), b.data_ptr(), out.data_ptr(),
                  bias=bias.data_ptr(), M=M, N=N, K=K, epi=EPI_BIAS_GELU)
    ref = torch.nn.functional.gelu(a.float() @ b.float().t() + bias,
                                   approximate="tanh")
    check(out, ref)


def test_gemm_bt_bf16(C):
    M, N, K = 256, 256, 128
    g = torch.Generator(device="cuda").manual_seed(9)
    a = torch.randn(M, K, generator=g, device="cuda").bfloat16().contiguous()
    b = torch.randn(N, K, generator=g, device="cuda").bfloat16().contiguous()
    out = torch.empty(M, N, dtype=torch.bfloat16, device="cuda")
    torch.cuda.synchronize()
    C.ops.gemm_bt(1, a.data_ptr(), b.data_ptr(), out.data_ptr(),
                  M=M, N=N, K=K, epi=EPI_NONE)
    ref = a.float() @ b.float().t()
    check(out, ref, rtol=4e-2, atol=4e-2)


def _conv_ref(x, w, stride, pad, scale=None, bias=None, res=None, epi=EPI_NONE):
    xc = x.float().permute(0, 3, 1, 2)
    wc = w.float()  # [Cout, Cin, KH, KW]
    y = torch.nn.functional.conv2d(xc, wc, stride=stride, padding=pad)
    y = y.permute(0, 2, 3, 1)
    if epi in (EPI_SB, EPI_SB_RELU, EPI_SB_ADD_RELU):
        y = y * scale + bias
    if epi == EPI_SB_ADD_RELU:
        y = y + res.float()
    if epi in (EPI_SB_RELU, EPI_SB_ADD_RELU):
        y = torch.relu(y)
    return y


def _pack_w(w):
    # [Cout, Cin, KH, KW] -> [Cout][KH*KW*Cin] padded to K%64
    cout, cin, kh, kw = w.shape
    flat = w.permute(0, 2, 3, 1).reshape(cout, kh * kw * cin)
    kp = ((kh * kw * cin + 63) // 64) * 64
    if kp != flat.shape[1]:
        flat = torch.nn.functional.pad(flat, (0, kp - flat.shape[1]))
    return flat.half().contiguous()


@pytest.mark.parametrize("shape", [
    # (Nb, H, W, Cin, Cout, KH, stride, pad)
    (2, 16, 16, 64, 128, 1, 1, 0),       # 1x1
    (2, 16, 16, 64, 64, 3, 1, 1),        # 3x3 s1
    (2, 32, 32, 64, 128, 3, 2, 1),       # 3x3 s2
    (2, 16, 16, 128, 64, 1, 2, 0),       # 1x1 s2 downsample
    (1, 56, 56, 8, 64, 7, 2, 3),         # stem-like 7x7 (C=8 padded)
])
def test_conv2d(C, shape):
    nb, h, w_, cin, cout, k, s, p = shape
    x = t16(nb, h, w_, cin, seed=sum(shape))
    wt = t16(cout, cin, k, k, seed=sum(shape) + 1, scale=0.1)
    zero = torch.zeros(64, dtype=torch.half, device="cuda")
    oh = (h + 2 * p - k) // s + 1
    ow = (w_ + 2 * p - k) // s + 1
    out = torch.empty(nb, oh, ow, cout, dtype=torch.half, device="cuda")
    wp = _pack_w(wt)
    torch.cuda.synchronize()
    C.ops.conv2d(0, x.data_ptr(), wp.data_ptr(), out.data_ptr(),
                 zero_page=zero.data_ptr(), Nb=nb, H=h, W=w_, C=cin,
                 Cout=cout, KH=k, KW=k, sh=s, sw=s, ph=p, pw=p, epi=EPI_NONE)
    ref = _conv_ref(x, wt, s, p)
    check(out, ref)


def test_conv2d_fused_bn_add_relu(C):
    nb, h, w_, cin, cout = 2, 14, 14, 256, 256
    x = t16(nb, h, w_, cin, seed=42)
    wt = t16(cout, cin, 3, 3, seed=43, scale=0.05)
    scale = torch.rand(cout, device="cuda").float() + 0.5
    bias = torch.randn(cout, device="cuda").float() * 0.1
    res = t16(nb, h, w_, cout, seed=44)
    zero = torch.zeros(64, dtype=torch.half, device="cuda")
    out = torch.empty(nb, h, w_, cout, dtype=torch.half, device="cuda")
    wp = _pack_w(wt)
    torch.cuda.synchronize()
    C.ops.conv2d(0, x.data_ptr(), wp.data_ptr(), out.data_ptr(),
                 scale=scale.data_ptr(), bias=bias.data_ptr(),
                 residual=res.data_ptr(), zero_page=zero.data_ptr(),
                 Nb=nb, H=h, W=w_, C=cin, Cout=cout, KH=3, KW=3, sh=1, sw=1,
                 ph=1, pw=1, epi=EPI_SB_ADD_RELU)
    ref = _conv_ref(x, wt, 1, 1, scale, bias, res, EPI_SB_ADD_RELU)
    check(out, ref)


def test_maxpool(C):
    x = t16(2, 32, 32, 64, seed=5)
    out = torch.empty(2, 16, 16, 64, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.maxpool2d(0, x.data_ptr(), out.data_ptr(), 2, 32, 32, 64, 3, 3,
                    2, 2, 1, 1)
    ref = torch.nn.functional.max_pool2d(
        x.float().permute(0, 3, 1, 2), 3, stride=2, padding=1
    ).permute(0, 2, 3, 1)
    check(out, ref, rtol=1e-3, atol=1e-3)


def test_gavgpool(C):
    x = t16(4, 7, 7, 2048, seed=6)
    out = torch.empty(4, 2048, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.gavgpool(0, x.data_ptr(), out.data_ptr(), 4, 49, 2048)
    ref = x.float().reshape(4, 49, 2048).mean(dim=1)
    check(out, ref, rtol=1e-2, atol=1e-3)


def test_softmax_rows(C):
    x = t16(100, 1000, seed=7, scale=3.0)
    out = torch.empty_like(x)
    torch.cuda.synchronize()
    C.ops.softmax_rows(0, x.data_ptr(), out.data_ptr(), 100, 1000)
    ref = torch.softmax(x.float(), dim=-1)
    check(out, ref, rtol=1e-2, atol=1e-4)


def test_layernorm(C):
    M, N = 256, 768
    x = t16(M, N, seed=8, scale=2.0)
    g_ = (torch.rand(N, device="cuda") + 0.5).float()
    b_ = torch.randn(N, device="cuda").float()
    out = torch.empty_like(x)
    torch.cuda.synchronize()
    C.ops.layernorm(0, x.data_ptr(), g_.data_ptr(), b_.data_ptr(),
                    out.data_ptr(), M, N, 1e-5)
    ref = torch.nn.functional.layer_norm(x.float(), (N,), g_, b_, 1e-5)
    check(out, ref)


def test_add_layernorm(C):
    M, N = 256, 768
    x, r = t16(M, N, seed=9), t16(M, N, seed=10)
    g_ = (torch.rand(N, device="cuda") + 0.5).float()
    b_ = torch.randn(N, device="cuda").float()
    out = torch.empty_like(x)
    torch.cuda.synchronize()
    C.ops.add_layernorm(0, x.data_ptr(), r.data_ptr(), g_.data_ptr(),
                        b_.data_ptr(), out.data_ptr(), 0, M, N, 1e-5)
    ref = torch.nn.functional.layer_norm((x + r).float(), (N,), g_, b_, 1e-5)
    check(out, ref)


def test_elementwise_add_relu(C):
    a, b = t16(64, 128, seed=11), t16(64, 128, seed=12)
    out = torch.empty_like(a)
    torch.cuda.synchronize()
    C.ops.elementwise(0, 3, a.data_ptr(), b.data_ptr(), out.data_ptr(),
                      a.numel())
    ref = torch.relu(a.float() + b.float())
    check(out, ref, rtol=1e-3, atol=1e-3)


def test_channel_pad(C):
    x = t16(100, 3, seed=13)
    out = torch.empty(100, 8, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.channel_pad(0, x.data_ptr(), out.data_ptr(), 100, 3, 8)
    ref = torch.nn.functional.pad(x.float(), (0, 5))
    check(out, ref, rtol=1e-3, atol=1e-3)


def test_attention(C):
    B, S, H, D = 2, 128, 12, 64
    hid = H * D
    qkv = t16(B * S, 3 * hid, seed=77)
    out = torch.empty(B * S, hid, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.attention(0, qkv.data_ptr(), out.data_ptr(), B, S, H, D,
                    1.0 / (D ** 0.5))
    q = qkv.float().reshape(B, S, 3, H, D)
    att = torch.softmax(
        (q[:, :, 0].permute(0, 2, 1, 3) @ q[:, :, 1].permute(0, 2, 1, 3).transpose(-1, -2))
        / (D ** 0.5), dim=-1)
    ref = (att @ q[:, :, 2].permute(0, 2, 1, 3)).permute(0, 2, 1, 3).reshape(B * S, hid)
    check(out, ref)


def test_gemm_bt_int8(C):
    M, N, K = 256, 512, 384  # K % 128 == 0
    g = torch.Generator(device="cuda").manual_seed(21)
    a = torch.randint(-127, 128, (M, K), generator=g, device="cuda",
                      dtype=torch.int8)
    b = torch.randint(-127, 128, (N, K), generator=g, device="cuda",
                      dtype=torch.int8)
    scale = torch.full((N,), 1e-3, device="cuda").float()
    bias = torch.randn(N, generator=g, device="cuda").float()
    out = torch.empty(M, N, dtype=torch.int8, device="cuda")
    torch.cuda.synchronize()
    C.ops.gemm_bt(2, a.data_ptr(), b.data_ptr(), out.data_ptr(),
                  scale=scale.data_ptr(), bias=bias.data_ptr(),
                  M=M, N=N, K=K, epi=EPI_SB)
    acc = a.float() @ b.float().t()
    ref = torch.clamp(torch.round(acc * scale + bias), -127, 127)
    err = (out.float() - ref).abs()
    assert (err <= 1).all(), f"max int8 err {err.max()}"
    assert (err > 0).float().mean() < 1e-3


def test_conv2d_int8(C):
    nb, h, w_, cin, cout = 2, 14, 14, 256, 256
    g = torch.Generator(device="cuda").manual_seed(22)
    x = torch.randint(-64, 65, (nb, h, w_, cin), generator=g, device="cuda",
                      dtype=torch.int8)
    wt = torch.randint(-64, 65, (cout, cin, 3, 3), generator=g,
                       device="cuda", dtype=torch.int8)
    flat = wt.permute(0, 2, 3, 1).reshape(cout, 9 * cin)
    kp = ((9 * cin + 127) // 128) * 128
    if kp != flat.shape[1]:
        flat = torch.nn.functional.pad(flat, (0, kp - flat.shape[1]))
    wp = flat.contiguous()
    scale = torch.full((cout,), 2e-5, device="cuda").float()
    bias = torch.zeros(cout, device="cuda").float()
    zero = torch.zeros(128, dtype=torch.int8, device="cuda")
    out = torch.empty(nb, h, w_, cout, dtype=torch.int8, device="cuda")
    torch.cuda.synchronize()
    C.ops.conv2d(2, x.data_ptr(), wp.data_ptr(), out.data_ptr(),
                 scale=scale.data_ptr(), bias=bias.data_ptr(),
                 zero_page=zero.data_ptr(), Nb=nb, H=h, W=w_, C=cin,
                 Cout=cout, KH=3, KW=3, sh=1, sw=1, ph=1, pw=1, epi=EPI_SB)
    acc = torch.nn.functional.conv2d(x.float().permute(0, 3, 1, 2),
                                     wt.float(), padding=1).permute(0, 2, 3, 1)
    ref = torch.clamp(torch.round(acc * scale + bias), -127, 127)
    err = (out.float() - ref).abs()
    assert (err <= 1).all(), f"max int8 err {err.max()}"


def test_quantize_dequant(C):
    x = t16(64, 128, seed=30, scale=2.0)
    q = torch.empty(64, 128, dtype=torch.int8, device="cuda")
    torch.cuda.synchronize()
    C.ops.quantize(x.data_ptr(), q.data_ptr(), x.numel(), 0.05)
    ref_q = torch.clamp(torch.round(x.float() / 0.05), -127, 127)
    assert (q.float() - ref_q).abs().max() <= 1
    back = torch.empty(64, 128, dtype=torch.half, device="cuda")
    C.ops.dequant(q.data_ptr(), back.data_ptr(), q.numel(), 0.05)
    # fp16 output rounding: ~2^-11 relative at |x| ~ 6
    assert (back.float() - q.float() * 0.05).abs().max() < 4e-3


def test_gemm_bt_fp8(C):
    """fp8 e4m3 MFMA GEMM vs dequantized torch reference."""
    M, N, K = 256, 512, 384
    g = torch.Generator(device="cuda").manual_seed(31)
    af = torch.randn(M, K, generator=g, device="cuda")
    bf = torch.randn(N, K, generator=g, device="cuda")
    a8 = af.to(torch.float8_e4m3fn)
    b8 = bf.to(torch.float8_e4m3fn)
    scale = torch.full((N,), 1.0, device="cuda").float()
    bias = torch.zeros(N, device="cuda").float()
    out = torch.empty(M, N, dtype=torch.float8_e4m3fn, device="cuda")
    torch.cuda.synchronize()
    C.ops.gemm_bt(3, a8.data_ptr(), b8.data_ptr(), out.data_ptr(),
                  scale=scale.data_ptr(), bias=bias.data_ptr(),
                  M=M, N=N, K=K, epi=EPI_SB)
    acc = a8.float() @ b8.float().t()
    ref = torch.clamp(acc, -448, 448).to(torch.float8_e4m3fn).float()
    err = (out.float() - ref).abs()
    tol = 0.07 * ref.abs() + 0.6  # one e4m3 ulp on the requantized output
    bad = (err > tol).float().mean().item()
    assert bad < 1e-3, (bad, float(err.max()))


def test_quantize_dequant_fp8(C):
    x = t16(64, 128, seed=32, scale=2.0)
    q = torch.empty(64, 128, dtype=torch.uint8, device="cuda")
    torch.cuda.synchronize()
    C.ops.quantize(x.data_ptr(), q.data_ptr(), x.numel(), 0.05, fmt=1)
    ref = torch.clamp(x.float() / 0.05, -448, 448).to(torch.float8_e4m3fn)
    assert (q.view(torch.float8_e4m3fn).float() - ref.float()).abs().max() < 1e-3
    back = torch.empty(64, 128, dtype=torch.half, device="cuda")
    C.ops.dequant(q.data_ptr(), back.data_ptr(), q.numel(), 0.05, fmt=1)
    assert (back.float() - ref.float() * 0.05).abs().max() < 0.05


def test_avgpool2d(C):
    x = t16(2, 16, 16, 64, seed=40)
    out = torch.empty(2, 8, 8, 64, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.avgpool2d(0, x.data_ptr(), out.data_ptr(), 2, 16, 16, 64, 3, 3,
                    2, 2, 1, 1)
    ref = torch.nn.functional.avg_pool2d(
        x.float().permute(0, 3, 1, 2), 3, stride=2, padding=1,
        count_include_pad=False).permute(0, 2, 3, 1)
    check(out, ref)


@pytest.mark.parametrize("M,N,K", [(128, 128, 128), (256, 384, 256),
                                   (1000, 768, 768), (100, 64, 1024),
                                   (2048, 2048, 512)])  # last: 128-tile path
def test_gemm_mxfp8(C, M, N, K):
    """MXFP8 GEMM (scaled MFMA 16x16x128, e8m0 block scales) vs the fp32
    matmul of the dequantized operands (exact oracle: the kernel's math is
    fp8*2^e with fp32 accumulation, bit-reproducible on dequantized fp32
    for these magnitudes)."""
    from trtlab_amd.engine.mx import dequantize_mxfp8, quantize_mxfp8

    rng = np.random.RandomState(M + N + K)
    # block-varying magnitudes so the e8m0 scales actually differ
    a32 = (rng.randn(M, K) * np.exp(rng.randn(M, 1))).astype(np.float32)
    b32 = (rng.randn(N, K) * np.exp(rng.randn(N, 1))).astype(np.float32)
    aq, asc = quantize_mxfp8(a32)
    bq, bsc = quantize_mxfp8(b32)

    a = torch.from_numpy(aq).cuda()
    b = torch.from_numpy(bq).cuda()
    sa = torch.from_numpy(asc).cuda()
    sb = torch.from_numpy(bsc).cuda()
    out = torch.empty(M, N, dtype=torch.float32, device="cuda")
    C.ops.gemm_mxfp8(a.data_ptr(), b.data_ptr(), sa.data_ptr(), sb.data_ptr(),
                     out.data_ptr(), M, N, K)
    ref = torch.from_numpy(dequantize_mxfp8(aq, asc)) @ \
        torch.from_numpy(dequantize_mxfp8(bq, bsc)).t()
    check(out, ref.cuda(), rtol=1e-2, atol=1e-2)
    # and it must track the unquantized fp32 product closely (MX quality)
    full = torch.from_numpy(a32) @ torch.from_numpy(b32).t()
    corr = np.corrcoef(out.cpu().numpy().ravel(), full.numpy().ravel())[0, 1]
    assert corr > 0.99, corr


@pytest.mark.parametrize("M,N,K", [(128, 128, 256), (256, 384, 512),
                                   (1000, 768, 768),
                                   (2048, 2048, 512)])  # last: 128-tile path
def test_gemm_mxfp4(C, M, N, K):
    """MXFP4 GEMM (scaled MFMA, cbsz=4) vs the dequantized-fp32 oracle."""
    from trtlab_amd.engine.mx import dequantize_mxfp4, quantize_mxfp4

    rng = np.random.RandomState(M + N + K)
    a32 = (rng.randn(M, K) * np.exp(rng.randn(M, 1))).astype(np.float32)
    b32 = (rng.randn(N, K) * np.exp(rng.randn(N, 1))).astype(np.float32)
    aq, asc = quantize_mxfp4(a32)
    bq, bsc = quantize_mxfp4(b32)
    a = torch.from_numpy(aq).cuda()
    b = torch.from_numpy(bq).cuda()
    sa = torch.from_numpy(asc).cuda()
    sb = torch.from_numpy(bsc).cuda()
    out = torch.empty(M, N, dtype=torch.float32, device="cuda")
    C.ops.gemm_mxfp4(a.data_ptr(), b.data_ptr(), sa.data_ptr(), sb.data_ptr(),
                     out.data_ptr(), M, N, K)
    ref = torch.from_numpy(dequantize_mxfp4(aq, asc)) @ \
        torch.from_numpy(dequantize_mxfp4(bq, bsc)).t()
    check(out, ref.cuda(), rtol=1e-2, atol=1e-2)
    full = torch.from_numpy(a32) @ torch.from_numpy(b32).t()
    corr = np.corrcoef(out.cpu().numpy().ravel(), full.numpy().ravel())[0, 1]
    assert corr > 0.95, corr  # fp4: coarser grid than fp8


@pytest.mark.parametrize("S,causal,D", [
    (128, 0, 64), (256, 0, 64), (512, 0, 64), (128, 1, 64), (256, 1, 64),
    # arbitrary S: tail query blocks + masked tail key tiles
    (200, 0, 64), (100, 1, 64), (77, 0, 64), (384, 0, 64),
    # head_dim 128 variant (64-key LDS tiles)
    (128, 0, 128), (256, 1, 128), (200, 0, 128), (77, 0, 128),
])
def test_attention_long_and_causal(C, S, causal, D):
    """Online-softmax attention: streamed key tiles at any S (incl. non-
    multiples of the tile sizes), decoder-style causal mask, and both head
    dims {64, 128}, vs plain torch attention."""
    B, H = 2, 4
    hid = H * D
    qkv = t16(B * S, 3 * hid, seed=100 + S + causal + D)
    out = torch.empty(B * S, hid, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.attention(0, qkv.data_ptr(), out.data_ptr(), B, S, H, D,
                    1.0 / (D ** 0.5), causal=causal)
    q = qkv.float().reshape(B, S, 3, H, D)
    scores = (q[:, :, 0].permute(0, 2, 1, 3) @
              q[:, :, 1].permute(0, 2, 1, 3).transpose(-1, -2)) / (D ** 0.5)
    if causal:
        cm = torch.arange(S, device="cuda")[None, :] > \
            torch.arange(S, device="cuda")[:, None]
        scores = scores.masked_fill(cm[None, None], float("-inf"))
    att = torch.softmax(scores, dim=-1)
    ref = (att @ q[:, :, 2].permute(0, 2, 1, 3)).permute(
        0, 2, 1, 3).reshape(B * S, hid)
    check(out, ref)


@pytest.mark.parametrize("S,D", [(200, 64), (200, 128), (384, 128)])
def test_attention_varlen_odd_s(C, S, D):
    """Variable-length key masking (seqlens) combined with arbitrary S and
    both head dims: rows of each sequence attend only its valid keys."""
    B, H = 2, 4
    hid = H * D
    qkv = t16(B * S, 3 * hid, seed=300 + S + D)
    lens = torch.tensor([S, S * 2 // 3], dtype=torch.int32, device="cuda")
    out = torch.empty(B * S, hid, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.attention(0, qkv.data_ptr(), out.data_ptr(), B, S, H, D,
                    1.0 / (D ** 0.5), seqlens=lens.data_ptr())
    q = qkv.float().reshape(B, S, 3, H, D)
    scores = (q[:, :, 0].permute(0, 2, 1, 3) @
              q[:, :, 1].permute(0, 2, 1, 3).transpose(-1, -2)) / (D ** 0.5)
    for b in range(B):
        scores[b, :, :, lens[b]:] = float("-inf")
    att = torch.softmax(scores, dim=-1)
    ref = (att @ q[:, :, 2].permute(0, 2, 1, 3)).permute(
        0, 2, 1, 3).reshape(B * S, hid)
    # compare only valid query rows (padded rows are don't-care)
    valid = torch.zeros(B * S, dtype=torch.bool)
    for b in range(B):
        valid[b * S:b * S + int(lens[b])] = True
    check(out[valid.to(out.device)], ref[valid.to(ref.device)])


def test_gemm_bt_tile256(C):
    """The 256x128 GEMM tactic (code 5) vs torch on a transformer shape."""
    M, N, K = 2048, 768, 768
    a = t16(M, K, seed=61)
    b = t16(N, K, seed=62)
    out = torch.empty(M, N, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.gemm_bt(0, a.data_ptr(), b.data_ptr(), out.data_ptr(),
                  M=M, N=N, K=K, tile=5)
    check(out, (a.float() @ b.float().t()).half())


def test_gemm_bt_strided_operands(C):
    """Strided lda/ldb/ldc (sub-matrix views): C view[M,N] inside a larger
    buffer = A view @ B view^T, vs torch on the same views."""
    LDA, LDB, LDC = 1024, 896, 1536
    M, N, K = 192, 256, 512
    a_full = t16(M, LDA, seed=71)
    b_full = t16(N, LDB, seed=72)
    c_full = torch.zeros(M, LDC, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.gemm_bt(0, a_full.data_ptr(), b_full.data_ptr(),
                  c_full.data_ptr(), M=M, N=N, K=K,
                  lda=LDA, ldb=LDB, ldc=LDC)
    ref = a_full[:, :K].float() @ b_full[:, :K].float().t()
    check(c_full[:, :N], ref)
    # untouched tail of each C row stays zero
    assert (c_full[:, N:] == 0).all()


def test_clip_kernel(C):
    x = t16(64, 256, seed=91, scale=3.0)
    out = torch.empty_like(x)
    torch.cuda.synchronize()
    C.ops.clip(0, x.data_ptr(), out.data_ptr(), x.numel(), -1.5, 2.0)
    ref = torch.clamp(x.float(), -1.5, 2.0)
    check(out, ref, rtol=1e-3, atol=1e-3)


@pytest.mark.parametrize("M,N", [(64, 64), (100, 200), (1000, 768), (65, 1)])
def test_transpose2d_kernel(C, M, N):
    x = t16(M, N, seed=92)
    out = torch.empty(N, M, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.transpose2d(0, x.data_ptr(), out.data_ptr(), M, N)
    assert torch.equal(out, x.t().contiguous())


def test_copy2d_concat(C):
    a = t16(128, 96, seed=93)
    b = t16(128, 160, seed=94)
    out = torch.zeros(128, 256, dtype=torch.half, device="cuda")
    torch.cuda.synchronize()
    C.ops.copy2d(0, a.data_ptr(), out.data_ptr(), 128, 96, 256, 0)
    C.ops.copy2d(0, b.data_ptr(), out.data_ptr(), 128, 160, 256, 96)
    ref = torch.cat([a, b], dim=1)
    assert torch.equal(out, ref)


def test_gumbel_argmax_sampling():
    """Device-side Gumbel-max categorical sampling: temp<=0 == argmax;
    fixed (seed, pos) reproducible; empirical frequencies over many
    independent rows match softmax(logits/T)."""
    import torch

    import trtlab_amd

    C = trtlab_amd.native()
    V = 8
    logits_row = np.log(np.array([1, 2, 4, 8, 1, 1, 1, 2], np.float64))
    probs = np.exp(logits_row) / np.exp(logits_row).sum()
    M = 4096
    x = torch.from_numpy(np.tile(logits_row, (M, 1))).half().cuda()
    out = torch.zeros(M, dtype=torch.int32, device="cuda")
    temps = torch.full((M,), 1.0, dtype=torch.float32, device="cuda")
    seeds = torch.arange(M, dtype=torch.int32, device="cuda")
    pos = torch.zeros(M, dtype=torch.int32, device="cuda")

    # temp=0 -> plain argmax on every row
    t0 = torch.zeros(M, dtype=torch.float32, device="cuda")
    C.ops.gumbel_argmax_rows(x.data_ptr(), out.data_ptr(),
                             temps=t0.data_ptr(), seeds=seeds.data_ptr(),
                             pos=pos.data_ptr(), M=M, V=V)
    assert (out.cpu().numpy() == 3).all()  # index of the max (8)

    # T=1: 4096 independent draws (one per row seed) ~ softmax
    C.ops.gumbel_argmax_rows(x.data_ptr(), out.data_ptr(),
                             temps=temps.data_ptr(), seeds=seeds.data_ptr(),
                             pos=pos.data_ptr(), M=M, V=V)
    draws = out.cpu().numpy()
    freq = np.bincount(draws, minlength=V) / M
    assert np.abs(freq - probs).max() < 0.03, (freq, probs)

    # reproducible for fixed (seed, pos); different pos changes draws
    a = draws.copy()
    C.ops.gumbel_argmax_rows(x.data_ptr(), out.data_ptr(),
                             temps=temps.data_ptr(), seeds=seeds.data_ptr(),
                             pos=pos.data_ptr(), M=M, V=V)
    assert (out.cpu().numpy() == a).all()
    pos2 = torch.full((M,), 7, dtype=torch.int32, device="cuda")
    C.ops.gumbel_argmax_rows(x.data_ptr(), out.data_ptr(),
                             temps=temps.data_ptr(), seeds=seeds.data_ptr(),
                             pos=pos2.data_ptr(), M=M, V=V)
    assert (out.cpu().numpy() != a).mean() > 0.3  # fresh noise


def test_session_sample_tokens():
    """DecodeSession.sample_tokens: greedy rows equal step's argmax ids;
    sampled rows are reproducible per (seed, position)."""
    from trtlab_amd.engine.decode import DecodeSession
    from trtlab_amd.models import build_llama

    g = build_llama(batch=2, seq=32, hidden=512, layers=1, heads=4,
                    seed=0, vocab=1000)
    s = DecodeSession(g, batch=2, smax=32, capture=False, lm_head=True)
    gids = s.step(np.array([5, 9], np.int32), return_ids=True)
    greedy = s.sample_tokens(np.zeros(2, np.float32),
                             np.zeros(2, np.int32))
    np.testing.assert_array_equal(gids, greedy)
    t = np.full(2, 1.5, np.float32)
    sd = np.array([7, 8], np.int32)
    a = s.sample_tokens(t, sd)
    b = s.sample_tokens(t, sd)
    np.testing.assert_array_equal(a, b)  # same (seed, pos) -> same draw
    s.close()
