// trtlab_amd — MFMA GEMM for gfx950 (CDNA4), fp16/bf16 in, fp32 accumulate.
//
// C[M][N] = epilogue( A[M][K] @ B[N][K]^T )
//
// B is stored transposed ([N][K], "bt") — weights are prepacked at engine
// build time into this layout so both LDS tiles stage identically and every
// MFMA fragment read is a contiguous ds_read_b128.
//
// Structure (cdna_hip_programming.md §5, "minimum 2-phase" recipe):
//   - 128x128 output tile, BK=64, 4 waves (2x2), each wave a 64x64 sub-tile
//     as 4x4 fragments of v_mfma_f32_16x16x32_{f16,bf16}.
//   - global->LDS staging via __builtin_amdgcn_global_load_lds (16 B/lane),
//     double-buffered; one vmcnt(0) + barrier per K-tile.
//   - LDS XOR swizzle ((row&7)<<4 on the byte offset) applied on the SOURCE
//     address (glds writes lane-linear) and re-applied on every ds_read_b128
//     (both-sides-or-neither, §5.4 rule 21).
//   - XCD-aware bijective blockIdx swizzle (T1).
//
// Replaces the TensorRT-internal GEMM path of the reference
// (trtlab/tensorrt/src/workspace.cc:47 enqueueV2 — opaque engine kernels).
#include "gemm_common.h"

namespace trtlab {

template <typename T, Epi E>
__global__ __launch_bounds__(256) void gemm_bt_kernel(
    const T* __restrict__ A, const T* __restrict__ B, T* __restrict__ C,
    const float* __restrict__ scale, const float* __restrict__ bias,
    const T* __restrict__ residual, int M, int N, int K, int64_t lda,
    int64_t ldb, int64_t ldc, int tiles_n) {
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

  const int ktiles = K >> 6;  // K % 64 == 0 (host asserts)

  stage_tile_128x64<T>(A + (int64_t)m0 * lda, lda, m0, M, lds0, tid);
  stage_tile_128x64<T>(B + (int64_t)n0 * ldb, ldb, n0, N, lds0 + 16384, tid);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int t = 0; t < ktiles; ++t) {
    if (t + 1 < ktiles) {
      uint32_t nb = lds0 + (cur ^ 1) * 32768;
      stage_tile_128x64<T>(A + (int64_t)m0 * lda + (t + 1) * 64, lda, m0, M,
                           nb, tid);
      stage_tile_128x64<T>(B + (int64_t)n0 * ldb + (t + 1) * 64, ldb, n0, N,
                           nb + 16384, tid);
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

  // Epilogue. D mapping for 16x16x32: col = lane&15, row = (lane>>4)*4 + r.
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wc * 64 + j * 16 + (lane & 15);
      if (col >= N) continue;
      float sc = 1.0f, bi = 0.0f;
      if constexpr (E == Epi::kScaleBias || E == Epi::kScaleBiasRelu ||
                    E == Epi::kScaleBiasAddRelu)
        sc = scale[col];
      if constexpr (E != Epi::kNone) bi = bias[col];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + i * 16 + ((lane >> 4) << 2) + r;
        if (row >= M) continue;
        float res = 0.0f;
        if constexpr (E == Epi::kScaleBiasAddRelu)
          res = (float)residual[(int64_t)row * ldc + col];
        float v = apply_epi<E>(acc[i][j][r], sc, bi, res);
        C[(int64_t)row * ldc + col] = (T)v;
      }
    }
  }
}

template <typename T>
static void launch_gemm_bt_t(const void* A, const void* B, void* C,
                             const float* scale, const float* bias,
                             const void* residual, int M, int N, int K,
                             int64_t lda, int64_t ldb, int64_t ldc, int epi,
                             hipStream_t stream) {
  int tiles_m = (int)cdiv(M, 128);
  int tiles_n = (int)cdiv(N, 128);
  dim3 grid(tiles_m * tiles_n);
  dim3 block(256);
  epi_dispatch(epi, [&](auto e) {
    constexpr Epi EE = decltype(e)::value;
    hipLaunchKernelGGL((gemm_bt_kernel<T, EE>), grid, block, 0, stream,
                       (const T*)A, (const T*)B, (T*)C, scale, bias,
                       (const T*)residual, M, N, K, lda, ldb, ldc, tiles_n);
  });
}

void launch_gemm_bt(int dtype,  // 0 = fp16, 1 = bf16
                    const void* A, const void* B, void* C, const float* scale,
                    const float* bias, const void* residual, int M, int N,
                    int K, int64_t lda, int64_t ldb, int64_t ldc, int epi,
                    hipStream_t stream) {
  if (K % 64 != 0) throw std::runtime_error("gemm_bt: K must be a multiple of 64");
  if (dtype == 0)
    launch_gemm_bt_t<_Float16>(A, B, C, scale, bias, residual, M, N, K, lda,
                               ldb, ldc, epi, stream);
  else
    launch_gemm_bt_t<__bf16>(A, B, C, scale, bias, residual, M, N, K, lda, ldb,
                             ldc, epi, stream);
}

}  // namespace trtlab
