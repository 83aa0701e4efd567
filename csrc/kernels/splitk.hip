// trtlab_amd — split-K reduction kernel (gfx950).
//
// Severely grid-starved K-heavy shapes (ResNet stage-4 3x3 at batch 8:
// 56 blocks on 256 CUs) split their K loop across `splitk` slices; each
// slice writes an fp32 [BM][BN] slab, and this kernel sums the slices and
// applies the fused epilogue. Deterministic (no atomics). Vectorized
// float4 slab reads, grid-strided over the whole slab set; bm*bn is a
// power of two so tile/element decomposition is shifts.
#include "gemm_common.h"

namespace trtlab {

template <typename T, Epi E>
__global__ void splitk_reduce_kernel(const float* __restrict__ scratch,
                                     T* __restrict__ C,
                                     const float* __restrict__ scale,
                                     const float* __restrict__ bias,
                                     const T* __restrict__ residual,
                                     float res_scale, int M, int N,
                                     int64_t ldc, int tiles_n, int splitk,
                                     int log_elems, int bn, int64_t total4) {
  // total4 = tiles * (bm*bn) / 4 vector groups of 4 consecutive elements.
  for (int64_t g4 = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       g4 < total4; g4 += (int64_t)gridDim.x * blockDim.x) {
    int64_t g = g4 * 4;
    int tile = (int)(g >> log_elems);
    int e = (int)(g & ((1 << log_elems) - 1));
    int64_t elems = (int64_t)1 << log_elems;
    const float4v* base =
        (const float4v*)(scratch + ((int64_t)tile * splitk) * elems) + (e >> 2);
    float4v v = *base;
    for (int s = 1; s < splitk; ++s) {
      float4v u = *(const float4v*)((const float*)base + s * elems);
      v.x += u.x; v.y += u.y; v.z += u.z; v.w += u.w;
    }
    int m0 = (tile / tiles_n) * ((int)elems / bn);
    int n0 = (tile % tiles_n) * bn;
    int row = m0 + e / bn;
    int col0 = n0 + e % bn;  // 4 consecutive cols (bn % 4 == 0)
    if (row >= M) continue;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = col0 + j;
      if (col >= N) continue;
      float sc = 1.0f, bi = 0.0f;
      if constexpr (E == Epi::kScaleBias || E == Epi::kScaleBiasRelu ||
                    E == Epi::kScaleBiasAddRelu || E == Epi::kScaleBiasGelu)
        sc = scale[col];
      if constexpr (E != Epi::kNone) bi = bias[col];
      float res = 0.0f;
      if constexpr (E == Epi::kScaleBiasAddRelu)
        res = (float)residual[(int64_t)row * ldc + col] * res_scale;
      float vv = j == 0 ? v.x : (j == 1 ? v.y : (j == 2 ? v.z : v.w));
      C[(int64_t)row * ldc + col] = store_cast<T>(apply_epi<E>(vv, sc, bi, res));
    }
  }
}

void launch_splitk_reduce(int dtype, const float* scratch, void* C,
                          const float* scale, const float* bias,
                          const void* residual, float res_scale, int M, int N,
                          int64_t ldc, int tiles_m, int tiles_n, int splitk,
                          int bm, int bn, int epi, hipStream_t stream) {
  int elems = bm * bn;  // power of two (64/128 x 64/128)
  int log_elems = 31 - __builtin_clz(elems);
  int64_t total4 = (int64_t)tiles_m * tiles_n * elems / 4;
  int blocks = (int)std::min<int64_t>(cdiv(total4, 256), 2048);
  dim3 grid(blocks);
  dim3 block(256);
  epi_dispatch(epi, [&](auto e) {
    constexpr Epi EE = decltype(e)::value;
    if (dtype == 0)
      hipLaunchKernelGGL((splitk_reduce_kernel<_Float16, EE>), grid, block, 0,
                         stream, scratch, (_Float16*)C, scale, bias,
                         (const _Float16*)residual, res_scale, M, N, ldc,
                         tiles_n, splitk, log_elems, bn, total4);
    else if (dtype == 1)
      hipLaunchKernelGGL((splitk_reduce_kernel<__bf16, EE>), grid, block, 0,
                         stream, scratch, (__bf16*)C, scale, bias,
                         (const __bf16*)residual, res_scale, M, N, ldc,
                         tiles_n, splitk, log_elems, bn, total4);
    else if (dtype == 2)
      hipLaunchKernelGGL((splitk_reduce_kernel<int8_t, EE>), grid, block, 0,
                         stream, scratch, (int8_t*)C, scale, bias,
                         (const int8_t*)residual, res_scale, M, N, ldc,
                         tiles_n, splitk, log_elems, bn, total4);
    else
      hipLaunchKernelGGL((splitk_reduce_kernel<__hip_fp8_e4m3, EE>), grid,
                         block, 0, stream, scratch, (__hip_fp8_e4m3*)C, scale,
                         bias, (const __hip_fp8_e4m3*)residual, res_scale, M,
                         N, ldc, tiles_n, splitk, log_elems, bn, total4);
  });
}

}  // namespace trtlab
