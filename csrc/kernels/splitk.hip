// trtlab_amd — split-K reduction kernel (gfx950).
//
// K-heavy, grid-starved shapes (ResNet stage-4 3x3 at batch 8: 56 blocks on
// 256 CUs; BERT ff2) split their K loop across `splitk` slices; each slice
// writes an fp32 [BM][BN] slab, and this kernel sums the slices and applies
// the fused epilogue. Deterministic (no atomics); slab traffic for these
// shapes is a few MB at 8 TB/s.
#include "gemm_common.h"

namespace trtlab {

template <typename T, Epi E>
__global__ void splitk_reduce_kernel(const float* __restrict__ scratch,
                                     T* __restrict__ C,
                                     const float* __restrict__ scale,
                                     const float* __restrict__ bias,
                                     const T* __restrict__ residual, int M,
                                     int N, int64_t ldc, int tiles_n,
                                     int splitk, int bm, int bn) {
  int tile = blockIdx.x;
  int m0 = (tile / tiles_n) * bm;
  int n0 = (tile % tiles_n) * bn;
  const float* base = scratch + (int64_t)tile * splitk * bm * bn;
  int elems = bm * bn;
  for (int e = threadIdx.x; e < elems; e += blockDim.x) {
    int row = m0 + e / bn;
    int col = n0 + e % bn;
    if (row >= M || col >= N) continue;
    float v = 0.f;
    for (int s = 0; s < splitk; ++s) v += base[(int64_t)s * elems + e];
    float sc = 1.0f, bi = 0.0f;
    if constexpr (E == Epi::kScaleBias || E == Epi::kScaleBiasRelu ||
                  E == Epi::kScaleBiasAddRelu)
      sc = scale[col];
    if constexpr (E != Epi::kNone) bi = bias[col];
    float res = 0.0f;
    if constexpr (E == Epi::kScaleBiasAddRelu)
      res = (float)residual[(int64_t)row * ldc + col];
    C[(int64_t)row * ldc + col] = (T)apply_epi<E>(v, sc, bi, res);
  }
}

void launch_splitk_reduce(int dtype, const float* scratch, void* C,
                          const float* scale, const float* bias,
                          const void* residual, int M, int N, int64_t ldc,
                          int tiles_m, int tiles_n, int splitk, int bm,
                          int bn, int epi, hipStream_t stream) {
  dim3 grid(tiles_m * tiles_n);
  dim3 block(256);
  epi_dispatch(epi, [&](auto e) {
    constexpr Epi EE = decltype(e)::value;
    if (dtype == 0)
      hipLaunchKernelGGL((splitk_reduce_kernel<_Float16, EE>), grid, block, 0,
                         stream, scratch, (_Float16*)C, scale, bias,
                         (const _Float16*)residual, M, N, ldc, tiles_n,
                         splitk, bm, bn);
    else
      hipLaunchKernelGGL((splitk_reduce_kernel<__bf16, EE>), grid, block, 0,
                         stream, scratch, (__bf16*)C, scale, bias,
                         (const __bf16*)residual, M, N, ldc, tiles_n, splitk,
                         bm, bn);
  });
}

}  // namespace trtlab
