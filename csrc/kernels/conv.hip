// trtlab_amd — implicit-GEMM conv2d (NHWC, fp16/bf16) for gfx950 with fused
// BN(scale,bias) + ReLU + optional residual-add epilogue.
//
// out[n,oh,ow,co] = epi( sum_{kh,kw,ci} in[n,ih,iw,ci] * w[co, kh,kw,ci] )
//     ih = oh*sh - ph + kh, iw = ow*sw - pw + kw
//
// GEMM view: A[M=N*OH*OW][K=KH*KW*C] is generated on the fly — each
// global_load_lds lane computes its own source address ((row,k) -> NHWC
// address) and out-of-image taps read a 16-byte zero page instead
// (glds has per-lane SOURCE addressing, so implicit im2col costs no
// separate pass and padding costs no branches in the MFMA loop).
// Weights are prepacked [Cout][KH*KW*C] ("bt" layout) at engine build time.
// Requires C % 8 == 0 (host pads input channels, e.g. RGB 3 -> 8) and
// (KH*KW*C) % 64 == 0 after host-side K padding of the weight tensor.
//
// Same 128x128x64 double-buffered MFMA structure as gemm.hip.
// Replaces TensorRT's internal conv kernels (reference has none of its own;
// SURVEY.md §2.8).
#include "gemm_common.h"

namespace trtlab {

struct ConvParams {
  int Nb, H, W, C;        // input NHWC (C already padded to %8)
  int Cout, KH, KW;       // weights [Cout][KH*KW*C]
  int OH, OW;             // output spatial
  int sh, sw, ph, pw;     // stride / padding
  int M;                  // Nb*OH*OW
  int K;                  // KH*KW*C, padded to %64 on the weight side
  int Kreal;              // KH*KW*C before padding (taps beyond read zero)
  FastDiv d_ohw, d_ow, d_c, d_kw;  // dividers: OH*OW, OW, C, KW
};

// Stage a 128-row x 64-elem A-tile of the implicit im2col matrix.
template <typename T>
__device__ __forceinline__ void stage_conv_a(
    const T* __restrict__ in, const T* __restrict__ zero_page,
    const ConvParams p, int m0, int k0, uint32_t lds_base, int tid) {
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    uint32_t pbyte = c * 4096 + tid * 16;
    uint32_t row = pbyte >> 7;
    uint32_t kb = (pbyte & 127) ^ ((row & 7) << 4);  // source-side swizzle
    int m = m0 + (int)row;
    if (m >= p.M) m = p.M - 1;
    uint32_t n = fdiv((uint32_t)m, p.d_ohw);
    uint32_t rem = (uint32_t)m - n * (uint32_t)(p.OH * p.OW);
    uint32_t oh = fdiv(rem, p.d_ow);
    uint32_t ow = rem - oh * (uint32_t)p.OW;
    int k = k0 + (int)(kb >> 1);  // element index along K (fp16/bf16: 2 B)
    const char* src;
    if (k >= p.Kreal) {
      src = (const char*)zero_page;
    } else {
      uint32_t pix = fdiv((uint32_t)k, p.d_c);
      uint32_t ci = (uint32_t)k - pix * (uint32_t)p.C;
      uint32_t kh = fdiv(pix, p.d_kw);
      uint32_t kw = pix - kh * (uint32_t)p.KW;
      int ih = (int)oh * p.sh - p.ph + (int)kh;
      int iw = (int)ow * p.sw - p.pw + (int)kw;
      if ((uint32_t)ih < (uint32_t)p.H && (uint32_t)iw < (uint32_t)p.W) {
        int64_t off =
            (((int64_t)n * p.H + ih) * p.W + iw) * p.C + ci;
        src = (const char*)(in + off);
      } else {
        src = (const char*)zero_page;
      }
    }
    glds16(src, lds_base + pbyte);
  }
}

template <typename T, Epi E>
__global__ __launch_bounds__(256) void conv_igemm_kernel(
    const T* __restrict__ in, const T* __restrict__ Wt, T* __restrict__ out,
    const float* __restrict__ scale, const float* __restrict__ bias,
    const T* __restrict__ residual, const T* __restrict__ zero_page,
    const ConvParams p, int tiles_n) {
  uint32_t bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int m0 = (int)(bid / tiles_n) * 128;
  int n0 = (int)(bid % tiles_n) * 128;

  __shared__ __attribute__((aligned(16))) char smem[2 * 2 * 16384];
  uint32_t lds0 = (uint32_t)(uintptr_t)&smem[0];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;

  using MF = Mfma16x16x32<T>;
  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int ktiles = p.K >> 6;

  stage_conv_a<T>(in, zero_page, p, m0, 0, lds0, tid);
  stage_tile_128x64<T>(Wt + (int64_t)n0 * p.K, p.K, n0, p.Cout, lds0 + 16384,
                       tid);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int t = 0; t < ktiles; ++t) {
    if (t + 1 < ktiles) {
      uint32_t nb = lds0 + (cur ^ 1) * 32768;
      stage_conv_a<T>(in, zero_page, p, m0, (t + 1) * 64, nb, tid);
      stage_tile_128x64<T>(Wt + (int64_t)n0 * p.K + (t + 1) * 64, p.K, n0,
                           p.Cout, nb + 16384, tid);
    }
    const char* As = &smem[cur * 32768];
    const char* Bs = As + 16384;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      typename MF::frag af[4], bf[4];
      uint32_t kbyte = ks * 64 + ((lane >> 4) << 4);
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        af[f] = read_frag<T>(As, wr * 64 + f * 16 + (lane & 15), kbyte);
        bf[f] = read_frag<T>(Bs, wc * 64 + f * 16 + (lane & 15), kbyte);
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = MF::run(af[i], bf[j], acc[i][j]);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wc * 64 + j * 16 + (lane & 15);
      if (col >= p.Cout) continue;
      float sc = 1.0f, bi = 0.0f;
      if constexpr (E == Epi::kScaleBias || E == Epi::kScaleBiasRelu ||
                    E == Epi::kScaleBiasAddRelu)
        sc = scale[col];
      if constexpr (E != Epi::kNone) bi = bias[col];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + i * 16 + ((lane >> 4) << 2) + r;
        if (row >= p.M) continue;
        float res = 0.0f;
        if constexpr (E == Epi::kScaleBiasAddRelu)
          res = (float)residual[(int64_t)row * p.Cout + col];
        float v = apply_epi<E>(acc[i][j][r], sc, bi, res);
        out[(int64_t)row * p.Cout + col] = (T)v;
      }
    }
  }
}

template <typename T>
static void launch_conv2d_t(const void* in, const void* Wt, void* out,
                            const float* scale, const float* bias,
                            const void* residual, const void* zero_page,
                            const ConvParams& p, int epi, hipStream_t stream) {
  int tiles_m = (int)cdiv(p.M, 128);
  int tiles_n = (int)cdiv(p.Cout, 128);
  dim3 grid(tiles_m * tiles_n);
  dim3 block(256);
  epi_dispatch(epi, [&](auto e) {
    constexpr Epi EE = decltype(e)::value;
    hipLaunchKernelGGL((conv_igemm_kernel<T, EE>), grid, block, 0, stream,
                       (const T*)in, (const T*)Wt, (T*)out, scale, bias,
                       (const T*)residual, (const T*)zero_page, p, tiles_n);
  });
}

void launch_conv2d(int dtype, const void* in, const void* Wt, void* out,
                   const float* scale, const float* bias, const void* residual,
                   const void* zero_page, int Nb, int H, int W, int C,
                   int Cout, int KH, int KW, int sh, int sw, int ph, int pw,
                   int epi, hipStream_t stream) {
  ConvParams p;
  p.Nb = Nb; p.H = H; p.W = W; p.C = C;
  p.Cout = Cout; p.KH = KH; p.KW = KW;
  p.sh = sh; p.sw = sw; p.ph = ph; p.pw = pw;
  p.OH = (H + 2 * ph - KH) / sh + 1;
  p.OW = (W + 2 * pw - KW) / sw + 1;
  p.M = Nb * p.OH * p.OW;
  p.Kreal = KH * KW * C;
  p.K = (int)round_up(p.Kreal, 64);
  if (C % 8 != 0) throw std::runtime_error("conv2d: C must be a multiple of 8 (pad input channels)");
  p.d_ohw = make_fastdiv((uint32_t)(p.OH * p.OW));
  p.d_ow = make_fastdiv((uint32_t)p.OW);
  p.d_c = make_fastdiv((uint32_t)C);
  p.d_kw = make_fastdiv((uint32_t)KW);
  if (dtype == 0)
    launch_conv2d_t<_Float16>(in, Wt, out, scale, bias, residual, zero_page, p,
                              epi, stream);
  else
    launch_conv2d_t<__bf16>(in, Wt, out, scale, bias, residual, zero_page, p,
                            epi, stream);
}

}  // namespace trtlab
