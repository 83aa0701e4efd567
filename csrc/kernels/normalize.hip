// trtlab_amd — row softmax and layernorm for gfx950 (wave64 shuffle
// reductions, fp32 accumulation, vectorized fp16/bf16 I/O).
// Covers SURVEY.md §2.8 items 6 (softmax) and 8 (layernorm).
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

// One wave per row; rows [M][ld], N valid columns; out = softmax(row).
// N <= 64*VMAX (VMAX=32 -> N<=2048 per row held in registers).
template <typename T>
__global__ void softmax_rows_kernel(const T* __restrict__ in,
                                    T* __restrict__ out, int M, int N,
                                    int64_t ld) {
  constexpr int VMAX = 32;
  int lane = threadIdx.x & 63;
  int row = (blockIdx.x * (blockDim.x >> 6)) + (threadIdx.x >> 6);
  if (row >= M) return;
  const T* src = in + (int64_t)row * ld;
  T* dst = out + (int64_t)row * ld;
  float v[VMAX];
  int cnt = 0;
  float m = -3.0e38f;
  for (int i = lane; i < N; i += 64) {
    v[cnt] = (float)src[i];
    m = fmaxf(m, v[cnt]);
    ++cnt;
  }
  m = wave_reduce_max(m);
  float s = 0.f;
  for (int c = 0; c < cnt; ++c) {
    v[c] = __expf(v[c] - m);
    s += v[c];
  }
  s = wave_reduce_sum(s);
  float inv = 1.0f / s;
  cnt = 0;
  for (int i = lane; i < N; i += 64) dst[i] = (T)(v[cnt++] * inv);
}

// One wave per row layernorm: out = (x - mean) / sqrt(var + eps) * gamma + beta
template <typename T>
__global__ void layernorm_kernel(const T* __restrict__ in,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ beta,
                                 T* __restrict__ out, int M, int N, int64_t ld,
                                 float eps) {
  constexpr int VMAX = 32;
  int lane = threadIdx.x & 63;
  int row = (blockIdx.x * (blockDim.x >> 6)) + (threadIdx.x >> 6);
  if (row >= M) return;
  const T* src = in + (int64_t)row * ld;
  T* dst = out + (int64_t)row * ld;
  float v[VMAX];
  int cnt = 0;
  float s = 0.f;
  for (int i = lane; i < N; i += 64) {
    v[cnt] = (float)src[i];
    s += v[cnt];
    ++cnt;
  }
  s = wave_reduce_sum(s);
  float mean = s / (float)N;
  float q = 0.f;
  for (int c = 0; c < cnt; ++c) {
    float d = v[c] - mean;
    q += d * d;
  }
  q = wave_reduce_sum(q);
  float rstd = rsqrtf(q / (float)N + eps);
  cnt = 0;
  for (int i = lane; i < N; i += 64) {
    float y = (v[cnt++] - mean) * rstd * gamma[i] + beta[i];
    dst[i] = (T)y;
  }
}

// Residual-add + layernorm fused (transformer block epilogue):
// out = LN(x + res), also writes the sum if sum_out != nullptr.
template <typename T>
__global__ void add_layernorm_kernel(const T* __restrict__ x,
                                     const T* __restrict__ res,
                                     const float* __restrict__ gamma,
                                     const float* __restrict__ beta,
                                     T* __restrict__ out, T* __restrict__ sum_out,
                                     int M, int N, int64_t ld, float eps) {
  constexpr int VMAX = 32;
  int lane = threadIdx.x & 63;
  int row = (blockIdx.x * (blockDim.x >> 6)) + (threadIdx.x >> 6);
  if (row >= M) return;
  const T* src = x + (int64_t)row * ld;
  const T* rsc = res + (int64_t)row * ld;
  T* dst = out + (int64_t)row * ld;
  float v[VMAX];
  int cnt = 0;
  float s = 0.f;
  for (int i = lane; i < N; i += 64) {
    v[cnt] = (float)src[i] + (float)rsc[i];
    s += v[cnt];
    ++cnt;
  }
  if (sum_out) {
    T* so = sum_out + (int64_t)row * ld;
    cnt = 0;
    for (int i = lane; i < N; i += 64) so[i] = (T)v[cnt++];
  }
  s = wave_reduce_sum(s);
  float mean = s / (float)N;
  float q = 0.f;
  for (int c = 0; c < cnt; ++c) {
    float d = v[c] - mean;
    q += d * d;
  }
  q = wave_reduce_sum(q);
  float rstd = rsqrtf(q / (float)N + eps);
  cnt = 0;
  for (int i = lane; i < N; i += 64) {
    float y = (v[cnt++] - mean) * rstd * gamma[i] + beta[i];
    dst[i] = (T)y;
  }
}

static inline dim3 rows_grid(int M) { return dim3((unsigned)cdiv(M, 4)); }

void launch_softmax_rows(int dtype, const void* in, void* out, int M, int N,
                         int64_t ld, hipStream_t stream) {
  if (N > 2048) throw std::runtime_error("softmax_rows: N > 2048 unsupported");
  if (dtype == 0)
    hipLaunchKernelGGL((softmax_rows_kernel<_Float16>), rows_grid(M), dim3(256),
                       0, stream, (const _Float16*)in, (_Float16*)out, M, N, ld);
  else
    hipLaunchKernelGGL((softmax_rows_kernel<__bf16>), rows_grid(M), dim3(256),
                       0, stream, (const __bf16*)in, (__bf16*)out, M, N, ld);
}

void launch_layernorm(int dtype, const void* in, const float* gamma,
                      const float* beta, void* out, int M, int N, int64_t ld,
                      float eps, hipStream_t stream) {
  if (N > 2048) throw std::runtime_error("layernorm: N > 2048 unsupported");
  if (dtype == 0)
    hipLaunchKernelGGL((layernorm_kernel<_Float16>), rows_grid(M), dim3(256), 0,
                       stream, (const _Float16*)in, gamma, beta, (_Float16*)out,
                       M, N, ld, eps);
  else
    hipLaunchKernelGGL((layernorm_kernel<__bf16>), rows_grid(M), dim3(256), 0,
                       stream, (const __bf16*)in, gamma, beta, (__bf16*)out, M,
                       N, ld, eps);
}

void launch_add_layernorm(int dtype, const void* x, const void* res,
                          const float* gamma, const float* beta, void* out,
                          void* sum_out, int M, int N, int64_t ld, float eps,
                          hipStream_t stream) {
  if (N > 2048) throw std::runtime_error("add_layernorm: N > 2048 unsupported");
  if (dtype == 0)
    hipLaunchKernelGGL((add_layernorm_kernel<_Float16>), rows_grid(M),
                       dim3(256), 0, stream, (const _Float16*)x,
                       (const _Float16*)res, gamma, beta, (_Float16*)out,
                       (_Float16*)sum_out, M, N, ld, eps);
  else
    hipLaunchKernelGGL((add_layernorm_kernel<__bf16>), rows_grid(M), dim3(256),
                       0, stream, (const __bf16*)x, (const __bf16*)res, gamma,
                       beta, (__bf16*)out, (__bf16*)sum_out, M, N, ld, eps);
}

}  // namespace trtlab
