// trtlab_amd — row softmax and layernorm for gfx950 (wave64 shuffle
// reductions, fp32 accumulation, vectorized 16-B fp16/bf16 I/O per G13:
// scalar bf16/fp16 loads cost ~2-2.5x on memory-bound kernels).
// Covers SURVEY.md §2.8 items 6 (softmax) and 8 (layernorm).
//
// Layout: one wave per row, 4 waves per block. Each lane owns 8-element
// chunks (chunk c = lane + 64*p); rows up to 2048 elements (N % 8 == 0).
#include <hip/hip_fp8.h>

#include "../common.h"

namespace trtlab {

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

template <typename T>
__device__ __forceinline__ void load8(const T* p, float* dst) {
  short8v raw = *(const short8v*)p;
#pragma unroll
  for (int j = 0; j < 8; ++j) dst[j] = (float)((const T*)&raw)[j];
}

template <typename T>
__device__ __forceinline__ void store8(T* p, const float* src) {
  T out[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = (T)src[j];
  *(short8v*)p = *(const short8v*)out;
}

constexpr int kMaxChunks = 4;  // 4 chunks/lane * 64 lanes * 8 elems = 2048

template <typename T>
__global__ void softmax_rows_kernel(const T* __restrict__ in,
                                    T* __restrict__ out, int M, int N,
                                    int64_t ld) {
  int lane = threadIdx.x & 63;
  int row = (blockIdx.x * (blockDim.x >> 6)) + (threadIdx.x >> 6);
  if (row >= M) return;
  const T* src = in + (int64_t)row * ld;
  T* dst = out + (int64_t)row * ld;
  float v[kMaxChunks][8];
  int nc = 0;
  float m = -3.0e38f;
  for (int c = lane; c * 8 < N; c += 64, ++nc) {
    load8(src + c * 8, v[nc]);
#pragma unroll
    for (int j = 0; j < 8; ++j) m = fmaxf(m, v[nc][j]);
  }
  m = wave_reduce_max(m);
  float s = 0.f;
  for (int q = 0; q < nc; ++q)
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      v[q][j] = __expf(v[q][j] - m);
      s += v[q][j];
    }
  s = wave_reduce_sum(s);
  float inv = 1.0f / s;
  nc = 0;
  for (int c = lane; c * 8 < N; c += 64, ++nc) {
#pragma unroll
    for (int j = 0; j < 8; ++j) v[nc][j] *= inv;
    store8(dst + c * 8, v[nc]);
  }
}

__device__ __forceinline__ void store8_fp8(unsigned char* p,
                                           const float* src,
                                           float inv_scale) {
  unsigned char out[8];
#pragma unroll
  for (int j = 0; j < 8; ++j)
    out[j] = __hip_fp8_e4m3(
                 fminf(fmaxf(src[j] * inv_scale, -448.f), 448.f)).__x;
  *(uint2*)p = *(const uint2*)out;
}

// Producer-fused OCP MX quantization of one normalized 8-elem chunk:
// the 32-element MX block spans the QUAD of adjacent lanes owning chunks
// 4q..4q+3, so the block amax is a 2-step shfl_xor quad-reduce. Code
// packing and e8m0 scale bias match quantize_mxfp4/8_kernel exactly
// (gemm_mx.hip) so the scaled-MFMA GEMMs consume either producer.
__device__ __forceinline__ void store8_mx(
    unsigned char* __restrict__ codes, unsigned char* __restrict__ scales,
    int chunk, const float* v8, int mx_mode /*4 or 8*/, int lane) {
  float amax = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) amax = fmaxf(amax, fabsf(v8[j]));
  amax = fmaxf(amax, __shfl_xor(amax, 1, 64));
  amax = fmaxf(amax, __shfl_xor(amax, 2, 64));
  int bias = (mx_mode == 4) ? 2 : 8;  // e2m1 emax 2, e4m3 emax 8
  int e = (amax > 0.f) ? (int)floorf(log2f(amax)) - bias : 0;
  e = e < -127 ? -127 : (e > 127 ? 127 : e);
  if ((lane & 3) == 0) scales[chunk >> 2] = (unsigned char)(e + 127);
  float inv = exp2f((float)-e);
  if (mx_mode == 8) {
    unsigned char out[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_fp8_e4m3 q = __hip_fp8_e4m3(
          fminf(fmaxf(v8[j] * inv, -448.f), 448.f));
      out[j] = *(const unsigned char*)&q;
    }
    *(float2v*)(codes + chunk * 8) = *(const float2v*)out;
  } else {
    unsigned char out[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      unsigned char byte = 0;
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        float q = v8[2 * j + h] * inv;
        float a = fabsf(q);
        int m;
        if (a < 0.25f) m = 0;
        else if (a < 0.75f) m = 1;
        else if (a < 1.25f) m = 2;
        else if (a < 1.75f) m = 3;
        else if (a < 2.5f) m = 4;
        else if (a < 3.5f) m = 5;
        else if (a < 5.0f) m = 6;
        else m = 7;
        unsigned char code = (unsigned char)(q < 0.f ? (m | 8) : m);
        byte |= (unsigned char)(code << (4 * h));
      }
      out[j] = byte;
    }
    *(float*)(codes + chunk * 4) = *(const float*)out;
  }
}

// layernorm core shared by the plain and residual-add variants. q_out, if
// set, receives the fp8-e4m3 quantized row (value / q_scale) — the fused
// producer-side quantization for fp8 transformer projections. mx_codes/
// mx_scales (mx_mode 4/8) instead emit the OCP MX row (codes + e8m0
// per-32-block scales) — producer-fused MX quantization.
template <typename T, bool ADD>
__device__ __forceinline__ void layernorm_row(
    const T* __restrict__ src, const T* __restrict__ res,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    T* __restrict__ dst, T* __restrict__ sum_out, int N, float eps, int lane,
    unsigned char* __restrict__ q_out = nullptr, float q_inv_scale = 1.0f,
    unsigned char* __restrict__ mx_codes = nullptr,
    unsigned char* __restrict__ mx_scales = nullptr, int mx_mode = 0) {
  float v[kMaxChunks][8];
  int nc = 0;
  float s = 0.f;
  for (int c = lane; c * 8 < N; c += 64, ++nc) {
    load8(src + c * 8, v[nc]);
    if constexpr (ADD) {
      float r[8];
      load8(res + c * 8, r);
#pragma unroll
      for (int j = 0; j < 8; ++j) v[nc][j] += r[j];
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) s += v[nc][j];
  }
  if (ADD && sum_out) {
    nc = 0;
    for (int c = lane; c * 8 < N; c += 64, ++nc) store8(sum_out + c * 8, v[nc]);
  }
  s = wave_reduce_sum(s);
  float mean = s / (float)N;
  float q = 0.f;
  for (int p = 0; p < nc; ++p)
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float d = v[p][j] - mean;
      q += d * d;
    }
  q = wave_reduce_sum(q);
  float rstd = rsqrtf(q / (float)N + eps);
  nc = 0;
  for (int c = lane; c * 8 < N; c += 64, ++nc) {
    float4v g0 = *(const float4v*)(gamma + c * 8);
    float4v g1 = *(const float4v*)(gamma + c * 8 + 4);
    float4v b0 = *(const float4v*)(beta + c * 8);
    float4v b1 = *(const float4v*)(beta + c * 8 + 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      v[nc][j] = (v[nc][j] - mean) * rstd * ((const float*)&g0)[j] +
                 ((const float*)&b0)[j];
      v[nc][4 + j] = (v[nc][4 + j] - mean) * rstd * ((const float*)&g1)[j] +
                     ((const float*)&b1)[j];
    }
    store8(dst + c * 8, v[nc]);
    if (q_out) store8_fp8(q_out + c * 8, v[nc], q_inv_scale);
    if (mx_codes) store8_mx(mx_codes, mx_scales, c, v[nc], mx_mode, lane);
  }
}

template <typename T>
__global__ void layernorm_kernel(const T* __restrict__ in,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ beta,
                                 T* __restrict__ out,
                                 unsigned char* __restrict__ q_out, int M,
                                 int N, int64_t ld, float eps,
                                 float q_inv_scale,
                                 unsigned char* __restrict__ mx_codes,
                                 unsigned char* __restrict__ mx_scales,
                                 int mx_mode) {
  int lane = threadIdx.x & 63;
  int row = (blockIdx.x * (blockDim.x >> 6)) + (threadIdx.x >> 6);
  if (row >= M) return;
  int64_t cstride = (mx_mode == 4) ? ld / 2 : ld;
  layernorm_row<T, false>(in + (int64_t)row * ld, nullptr, gamma, beta,
                          out + (int64_t)row * ld, nullptr, N, eps, lane,
                          q_out ? q_out + (int64_t)row * ld : nullptr,
                          q_inv_scale,
                          mx_codes ? mx_codes + (int64_t)row * cstride
                                   : nullptr,
                          mx_scales ? mx_scales + (int64_t)row * (ld / 32)
                                    : nullptr,
                          mx_mode);
}

template <typename T>
__global__ void add_layernorm_kernel(const T* __restrict__ x,
                                     const T* __restrict__ res,
                                     const float* __restrict__ gamma,
                                     const float* __restrict__ beta,
                                     T* __restrict__ out,
                                     T* __restrict__ sum_out,
                                     unsigned char* __restrict__ q_out, int M,
                                     int N, int64_t ld, float eps,
                                     float q_inv_scale,
                                     unsigned char* __restrict__ mx_codes,
                                     unsigned char* __restrict__ mx_scales,
                                     int mx_mode) {
  int lane = threadIdx.x & 63;
  int row = (blockIdx.x * (blockDim.x >> 6)) + (threadIdx.x >> 6);
  if (row >= M) return;
  int64_t cstride = (mx_mode == 4) ? ld / 2 : ld;
  layernorm_row<T, true>(x + (int64_t)row * ld, res + (int64_t)row * ld,
                         gamma, beta, out + (int64_t)row * ld,
                         sum_out ? sum_out + (int64_t)row * ld : nullptr, N,
                         eps, lane,
                         q_out ? q_out + (int64_t)row * ld : nullptr,
                         q_inv_scale,
                         mx_codes ? mx_codes + (int64_t)row * cstride
                                  : nullptr,
                         mx_scales ? mx_scales + (int64_t)row * (ld / 32)
                                   : nullptr,
                         mx_mode);
}

static inline dim3 rows_grid(int M) { return dim3((unsigned)cdiv(M, 4)); }

void launch_softmax_rows(int dtype, const void* in, void* out, int M, int N,
                         int64_t ld, hipStream_t stream) {
  if (N > 2048) throw std::runtime_error("softmax_rows: N > 2048 unsupported");
  if (N % 8 != 0) throw std::runtime_error("softmax_rows: N % 8 != 0");
  if (dtype == 0)
    hipLaunchKernelGGL((softmax_rows_kernel<_Float16>), rows_grid(M), dim3(256),
                       0, stream, (const _Float16*)in, (_Float16*)out, M, N, ld);
  else
    hipLaunchKernelGGL((softmax_rows_kernel<__bf16>), rows_grid(M), dim3(256),
                       0, stream, (const __bf16*)in, (__bf16*)out, M, N, ld);
}

void launch_layernorm(int dtype, const void* in, const float* gamma,
                      const float* beta, void* out, int M, int N, int64_t ld,
                      float eps, hipStream_t stream, void* q_out,
                      float q_scale, void* mx_codes, void* mx_scales,
                      int mx_mode) {
  if (N > 2048 || N % 8 != 0) throw std::runtime_error("layernorm: bad N");
  if (mx_mode && N % 32 != 0)
    throw std::runtime_error("layernorm mx: N % 32 != 0");
  float inv = q_scale != 0.f ? 1.0f / q_scale : 1.0f;
  if (dtype == 0)
    hipLaunchKernelGGL((layernorm_kernel<_Float16>), rows_grid(M), dim3(256), 0,
                       stream, (const _Float16*)in, gamma, beta, (_Float16*)out,
                       (unsigned char*)q_out, M, N, ld, eps, inv,
                       (unsigned char*)mx_codes, (unsigned char*)mx_scales,
                       mx_mode);
  else
    hipLaunchKernelGGL((layernorm_kernel<__bf16>), rows_grid(M), dim3(256), 0,
                       stream, (const __bf16*)in, gamma, beta, (__bf16*)out,
                       (unsigned char*)q_out, M, N, ld, eps, inv,
                       (unsigned char*)mx_codes, (unsigned char*)mx_scales,
                       mx_mode);
}

void launch_add_layernorm(int dtype, const void* x, const void* res,
                          const float* gamma, const float* beta, void* out,
                          void* sum_out, int M, int N, int64_t ld, float eps,
                          hipStream_t stream, void* q_out, float q_scale,
                          void* mx_codes, void* mx_scales, int mx_mode) {
  if (N > 2048 || N % 8 != 0) throw std::runtime_error("add_layernorm: bad N");
  if (mx_mode && N % 32 != 0)
    throw std::runtime_error("add_layernorm mx: N % 32 != 0");
  float inv = q_scale != 0.f ? 1.0f / q_scale : 1.0f;
  if (dtype == 0)
    hipLaunchKernelGGL((add_layernorm_kernel<_Float16>), rows_grid(M),
                       dim3(256), 0, stream, (const _Float16*)x,
                       (const _Float16*)res, gamma, beta, (_Float16*)out,
                       (_Float16*)sum_out, (unsigned char*)q_out, M, N, ld,
                       eps, inv, (unsigned char*)mx_codes,
                       (unsigned char*)mx_scales, mx_mode);
  else
    hipLaunchKernelGGL((add_layernorm_kernel<__bf16>), rows_grid(M), dim3(256),
                       0, stream, (const __bf16*)x, (const __bf16*)res, gamma,
                       beta, (__bf16*)out, (__bf16*)sum_out,
                       (unsigned char*)q_out, M, N, ld, eps, inv,
                       (unsigned char*)mx_codes, (unsigned char*)mx_scales,
                       mx_mode);
}


// ---- RMSNorm (LLaMA-family): out = x / rms(x) * gamma ----
// Same one-wave-per-row structure as layernorm; no mean subtraction, no
// beta (the LLaMA norm has none).
template <typename T>
__global__ void rmsnorm_kernel(const T* __restrict__ in,
                               const float* __restrict__ gamma,
                               T* __restrict__ out, int M, int N, int64_t ld,
                               float eps) {
  int lane = threadIdx.x & 63;
  int row = (blockIdx.x * (blockDim.x >> 6)) + (threadIdx.x >> 6);
  if (row >= M) return;
  const T* src = in + (int64_t)row * ld;
  T* dst = out + (int64_t)row * ld;
  float v[kMaxChunks][8];
  int nc = 0;
  float ss = 0.f;
  for (int c = lane; c * 8 < N; c += 64, ++nc) {
    load8(src + c * 8, v[nc]);
#pragma unroll
    for (int j = 0; j < 8; ++j) ss += v[nc][j] * v[nc][j];
  }
  ss = wave_reduce_sum(ss);
  float r = rsqrtf(ss / (float)N + eps);
  nc = 0;
  for (int c = lane; c * 8 < N; c += 64, ++nc) {
    float4v g0 = *(const float4v*)(gamma + c * 8);
    float4v g1 = *(const float4v*)(gamma + c * 8 + 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      v[nc][j] = v[nc][j] * r * ((const float*)&g0)[j];
      v[nc][4 + j] = v[nc][4 + j] * r * ((const float*)&g1)[j];
    }
    store8(dst + c * 8, v[nc]);
  }
}

// add + rmsnorm (residual stream variant; sum_out = updated residual)
template <typename T>
__global__ void add_rmsnorm_kernel(const T* __restrict__ x,
                                   const T* __restrict__ res,
                                   const float* __restrict__ gamma,
                                   T* __restrict__ out,
                                   T* __restrict__ sum_out, int M, int N,
                                   int64_t ld, float eps) {
  int lane = threadIdx.x & 63;
  int row = (blockIdx.x * (blockDim.x >> 6)) + (threadIdx.x >> 6);
  if (row >= M) return;
  const T* xs = x + (int64_t)row * ld;
  const T* rs = res + (int64_t)row * ld;
  T* dst = out + (int64_t)row * ld;
  T* sm = sum_out ? sum_out + (int64_t)row * ld : nullptr;
  float v[kMaxChunks][8];
  int nc = 0;
  float ss = 0.f;
  for (int c = lane; c * 8 < N; c += 64, ++nc) {
    float a[8], b[8];
    load8(xs + c * 8, a);
    load8(rs + c * 8, b);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      v[nc][j] = a[j] + b[j];
      ss += v[nc][j] * v[nc][j];
    }
    if (sm) store8(sm + c * 8, v[nc]);
  }
  ss = wave_reduce_sum(ss);
  float r = rsqrtf(ss / (float)N + eps);
  nc = 0;
  for (int c = lane; c * 8 < N; c += 64, ++nc) {
    float4v g0 = *(const float4v*)(gamma + c * 8);
    float4v g1 = *(const float4v*)(gamma + c * 8 + 4);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      v[nc][j] = v[nc][j] * r * ((const float*)&g0)[j];
      v[nc][4 + j] = v[nc][4 + j] * r * ((const float*)&g1)[j];
    }
    store8(dst + c * 8, v[nc]);
  }
}

void launch_rmsnorm(int dtype, const void* in, const float* gamma, void* out,
                    int M, int N, int64_t ld, float eps,
                    hipStream_t stream) {
  if (N > 2048 || N % 8 != 0) throw std::runtime_error("rmsnorm: bad N");
  if (dtype == 0)
    hipLaunchKernelGGL((rmsnorm_kernel<_Float16>), rows_grid(M), dim3(256),
                       0, stream, (const _Float16*)in, gamma,
                       (_Float16*)out, M, N, ld, eps);
  else
    hipLaunchKernelGGL((rmsnorm_kernel<__bf16>), rows_grid(M), dim3(256), 0,
                       stream, (const __bf16*)in, gamma, (__bf16*)out, M, N,
                       ld, eps);
}

void launch_add_rmsnorm(int dtype, const void* x, const void* res,
                        const float* gamma, void* out, void* sum_out, int M,
                        int N, int64_t ld, float eps, hipStream_t stream) {
  if (N > 2048 || N % 8 != 0) throw std::runtime_error("add_rmsnorm: bad N");
  if (dtype == 0)
    hipLaunchKernelGGL((add_rmsnorm_kernel<_Float16>), rows_grid(M),
                       dim3(256), 0, stream, (const _Float16*)x,
                       (const _Float16*)res, gamma, (_Float16*)out,
                       (_Float16*)sum_out, M, N, ld, eps);
  else
    hipLaunchKernelGGL((add_rmsnorm_kernel<__bf16>), rows_grid(M), dim3(256),
                       0, stream, (const __bf16*)x, (const __bf16*)res,
                       gamma, (__bf16*)out, (__bf16*)sum_out, M, N, ld, eps);
}

}  // namespace trtlab
