// trtlab_amd — elementwise kernels (gfx950): vectorized fp16/bf16, 16 B/lane.
// SURVEY.md §2.8 item 5; also dtype converts and the channel-pad copy used to
// bring NHWC inputs to C % 8 == 0 for the implicit-GEMM conv.
#include <hip/hip_fp8.h>

#include "../common.h"

namespace trtlab {

enum class EwOp : int { kRelu = 0, kGelu = 1, kAdd = 2, kAddRelu = 3 };

__device__ __forceinline__ float gelu_tanh_e(float x) {
  const float k0 = 0.7978845608028654f;
  const float k1 = 0.044715f;
  float u = k0 * (x + k1 * x * x * x);
  return 0.5f * x * (1.0f + tanhf(u));
}

template <typename T, EwOp OP>
__global__ void elementwise_kernel(const T* __restrict__ a,
                                   const T* __restrict__ b, T* __restrict__ o,
                                   int64_t n8) {
  // n8 = number of 8-element groups (callers pad buffers to 16 B)
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    short4v va0 = *(const short4v*)(a + i * 8);
    short4v va1 = *(const short4v*)(a + i * 8 + 4);
    T r[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float x = (float)((const T*)(j < 4 ? (const void*)&va0 : (const void*)&va1))[j & 3];
      float y;
      if constexpr (OP == EwOp::kRelu) y = fmaxf(x, 0.f);
      else if constexpr (OP == EwOp::kGelu) y = gelu_tanh_e(x);
      else {
        float xb = (float)b[i * 8 + j];
        y = x + xb;
        if constexpr (OP == EwOp::kAddRelu) y = fmaxf(y, 0.f);
      }
      r[j] = (T)y;
    }
    *(short4v*)(o + i * 8) = *(const short4v*)&r[0];
    *(short4v*)(o + i * 8 + 4) = *(const short4v*)&r[4];
  }
}

// NHWC channel pad: in [M, Cin] -> out [M, Cpad], zeros beyond Cin.
template <typename T>
__global__ void channel_pad_kernel(const T* __restrict__ in, T* __restrict__ out,
                                   int64_t M, int Cin, int Cpad) {
  int64_t total = M * Cpad;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c = (int)(i % Cpad);
    int64_t m = i / Cpad;
    out[i] = (c < Cin) ? in[m * Cin + c] : (T)0.0f;
  }
}

// Per-channel affine (+ optional ReLU): out[m, c] = x[m, c]*s[c] + b[c].
// Standalone batchnorm whose producer is not a conv (DenseNet's
// pre-activation BN after a concat; ONNX BatchNormalization on non-conv
// inputs) — the folded (scale, bias) come from the host. C % 8 == 0.
template <typename T, bool RELU>
__global__ void channel_affine_kernel(const T* __restrict__ x,
                                      T* __restrict__ out,
                                      const float* __restrict__ s,
                                      const float* __restrict__ b,
                                      int64_t n8, int C8) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    int c8 = (int)(i % C8) * 8;
    short4v v0 = *(const short4v*)(x + i * 8);
    short4v v1 = *(const short4v*)(x + i * 8 + 4);
    T r[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = (float)((const T*)(j < 4 ? (const void*)&v0
                                         : (const void*)&v1))[j & 3];
      v = v * s[c8 + j] + b[c8 + j];
      if constexpr (RELU) v = fmaxf(v, 0.0f);
      r[j] = (T)v;
    }
    *(short4v*)(out + i * 8) = *(const short4v*)&r[0];
    *(short4v*)(out + i * 8 + 4) = *(const short4v*)&r[4];
  }
}

// fp32 -> fp16/bf16 and back (bindings staging, tests)
template <typename T>
__global__ void cast_from_f32_kernel(const float* __restrict__ in,
                                     T* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (T)in[i];
}
template <typename T>
__global__ void cast_to_f32_kernel(const T* __restrict__ in,
                                   float* __restrict__ out, int64_t n) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (float)in[i];
}

static inline int ew_blocks(int64_t work) {
  return (int)std::min<int64_t>(cdiv(work, 256), 2048);
}

void launch_elementwise(int dtype, int op, const void* a, const void* b,
                        void* out, int64_t n, hipStream_t stream) {
  if (n % 8 != 0) throw std::runtime_error("elementwise: n % 8 != 0");
  int64_t n8 = n / 8;
  int blocks = ew_blocks(n8);
  auto l = [&](auto t, auto o) {
    using T = decltype(t);
    constexpr EwOp O = decltype(o)::value;
    hipLaunchKernelGGL((elementwise_kernel<T, O>), dim3(blocks), dim3(256), 0,
                       stream, (const T*)a, (const T*)b, (T*)out, n8);
  };
  auto dis = [&](auto t) {
    using T = decltype(t);
    switch ((EwOp)op) {
      case EwOp::kRelu: l(T{}, std::integral_constant<EwOp, EwOp::kRelu>{}); break;
      case EwOp::kGelu: l(T{}, std::integral_constant<EwOp, EwOp::kGelu>{}); break;
      case EwOp::kAdd: l(T{}, std::integral_constant<EwOp, EwOp::kAdd>{}); break;
      case EwOp::kAddRelu: l(T{}, std::integral_constant<EwOp, EwOp::kAddRelu>{}); break;
    }
  };
  if (dtype == 0) dis(_Float16{});
  else dis(__bf16{});
}

void launch_channel_affine(int dtype, const void* x, void* out,
                           const float* s, const float* b, int64_t M, int C,
                           bool relu, hipStream_t stream) {
  if (C % 8 != 0)
    throw std::runtime_error("channel_affine: C % 8 != 0");
  int64_t n8 = M * (int64_t)(C / 8);
  int blocks = ew_blocks(n8);
  auto l = [&](auto t, auto r) {
    using T = decltype(t);
    hipLaunchKernelGGL((channel_affine_kernel<T, decltype(r)::value>),
                       dim3(blocks), dim3(256), 0, stream, (const T*)x,
                       (T*)out, s, b, n8, C / 8);
  };
  if (dtype == 0) {
    relu ? l(_Float16{}, std::true_type{})
         : l(_Float16{}, std::false_type{});
  } else {
    relu ? l(__bf16{}, std::true_type{}) : l(__bf16{}, std::false_type{});
  }
}

void launch_channel_pad(int dtype, const void* in, void* out, int64_t M,
                        int Cin, int Cpad, hipStream_t stream) {
  int blocks = ew_blocks(M * Cpad);
  if (dtype == 0)
    hipLaunchKernelGGL((channel_pad_kernel<_Float16>), dim3(blocks), dim3(256),
                       0, stream, (const _Float16*)in, (_Float16*)out, M, Cin,
                       Cpad);
  else if (dtype == 1)
    hipLaunchKernelGGL((channel_pad_kernel<__bf16>), dim3(blocks), dim3(256),
                       0, stream, (const __bf16*)in, (__bf16*)out, M, Cin,
                       Cpad);
  else if (dtype == 2)
    hipLaunchKernelGGL((channel_pad_kernel<int8_t>), dim3(blocks), dim3(256),
                       0, stream, (const int8_t*)in, (int8_t*)out, M, Cin,
                       Cpad);
  else
    hipLaunchKernelGGL((channel_pad_kernel<__hip_fp8_e4m3>), dim3(blocks),
                       dim3(256), 0, stream, (const __hip_fp8_e4m3*)in,
                       (__hip_fp8_e4m3*)out, M, Cin, Cpad);
}

// fp16 -> int8 symmetric quantization (and inverse), vectorized 8-wide.
__global__ void quantize_kernel(const _Float16* __restrict__ in,
                                int8_t* __restrict__ out, int64_t n8,
                                float inv_scale) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    short4v v0 = *(const short4v*)(in + i * 8);
    short4v v1 = *(const short4v*)(in + i * 8 + 4);
    int8_t r[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float x = (float)((const _Float16*)(j < 4 ? (const void*)&v0
                                                : (const void*)&v1))[j & 3];
      float q = rintf(x * inv_scale);
      q = fminf(fmaxf(q, -127.f), 127.f);
      r[j] = (int8_t)q;
    }
    *(uint2*)(out + i * 8) = *(const uint2*)r;
  }
}

__global__ void dequant_kernel(const int8_t* __restrict__ in,
                               _Float16* __restrict__ out, int64_t n8,
                               float scale) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    int8_t r[8];
    *(uint2*)r = *(const uint2*)(in + i * 8);
    _Float16 o[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (_Float16)((float)r[j] * scale);
    *(short4v*)(out + i * 8) = *(const short4v*)&o[0];
    *(short4v*)(out + i * 8 + 4) = *(const short4v*)&o[4];
  }
}

// fp8 e4m3 variants (fmt = 1): continuous quantization, saturate at 448.
__global__ void quantize_fp8_kernel(const _Float16* __restrict__ in,
                                    unsigned char* __restrict__ out,
                                    int64_t n8, float inv_scale) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    short4v v0 = *(const short4v*)(in + i * 8);
    short4v v1 = *(const short4v*)(in + i * 8 + 4);
    unsigned char r[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float x = (float)((const _Float16*)(j < 4 ? (const void*)&v0
                                                : (const void*)&v1))[j & 3];
      float q = fminf(fmaxf(x * inv_scale, -448.f), 448.f);
      r[j] = __hip_fp8_e4m3(q).__x;
    }
    *(uint2*)(out + i * 8) = *(const uint2*)r;
  }
}

__global__ void dequant_fp8_kernel(const unsigned char* __restrict__ in,
                                   _Float16* __restrict__ out, int64_t n8,
                                   float scale) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    unsigned char r[8];
    *(uint2*)r = *(const uint2*)(in + i * 8);
    _Float16 o[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      __hip_fp8_e4m3 v;
      v.__x = r[j];
      o[j] = (_Float16)((float)v * scale);
    }
    *(short4v*)(out + i * 8) = *(const short4v*)&o[0];
    *(short4v*)(out + i * 8 + 4) = *(const short4v*)&o[4];
  }
}

void launch_quantize(const void* in_f16, void* out_q, int64_t n, float scale,
                     hipStream_t stream, int fmt) {
  if (n % 8 != 0) throw std::runtime_error("quantize: n % 8 != 0");
  int blocks = ew_blocks(n / 8);
  if (fmt == 0)
    hipLaunchKernelGGL(quantize_kernel, dim3(blocks), dim3(256), 0, stream,
                       (const _Float16*)in_f16, (int8_t*)out_q, n / 8,
                       1.0f / scale);
  else
    hipLaunchKernelGGL(quantize_fp8_kernel, dim3(blocks), dim3(256), 0,
                       stream, (const _Float16*)in_f16,
                       (unsigned char*)out_q, n / 8, 1.0f / scale);
}

void launch_dequant(const void* in_q, void* out_f16, int64_t n, float scale,
                    hipStream_t stream, int fmt) {
  if (n % 8 != 0) throw std::runtime_error("dequant: n % 8 != 0");
  int blocks = ew_blocks(n / 8);
  if (fmt == 0)
    hipLaunchKernelGGL(dequant_kernel, dim3(blocks), dim3(256), 0, stream,
                       (const int8_t*)in_q, (_Float16*)out_f16, n / 8, scale);
  else
    hipLaunchKernelGGL(dequant_fp8_kernel, dim3(blocks), dim3(256), 0, stream,
                       (const unsigned char*)in_q, (_Float16*)out_f16, n / 8,
                       scale);
}

void launch_cast(int dtype, bool to_f32, const void* in, void* out, int64_t n,
                 hipStream_t stream) {
  int blocks = ew_blocks(n);
  if (dtype == 0) {
    if (to_f32)
      hipLaunchKernelGGL((cast_to_f32_kernel<_Float16>), dim3(blocks),
                         dim3(256), 0, stream, (const _Float16*)in,
                         (float*)out, n);
    else
      hipLaunchKernelGGL((cast_from_f32_kernel<_Float16>), dim3(blocks),
                         dim3(256), 0, stream, (const float*)in,
                         (_Float16*)out, n);
  } else {
    if (to_f32)
      hipLaunchKernelGGL((cast_to_f32_kernel<__bf16>), dim3(blocks), dim3(256),
                         0, stream, (const __bf16*)in, (float*)out, n);
    else
      hipLaunchKernelGGL((cast_from_f32_kernel<__bf16>), dim3(blocks),
                         dim3(256), 0, stream, (const float*)in, (__bf16*)out,
                         n);
  }
}


// ---- general clip: out = min(max(x, mn), mx) (ONNX Clip with bounds) ----
template <typename T>
__global__ void clip_kernel(const T* __restrict__ in, T* __restrict__ out,
                            int64_t n8, float mn, float mx) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    short4v a0 = *(const short4v*)((const T*)in + i * 8);
    short4v a1 = *(const short4v*)((const T*)in + i * 8 + 4);
    T o[8];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      o[j] = (T)fminf(fmaxf((float)((const T*)&a0)[j], mn), mx);
      o[4 + j] = (T)fminf(fmaxf((float)((const T*)&a1)[j], mn), mx);
    }
    *(short4v*)(out + i * 8) = *(const short4v*)&o[0];
    *(short4v*)(out + i * 8 + 4) = *(const short4v*)&o[4];
  }
}

void launch_clip(int dtype, const void* in, void* out, int64_t n, float mn,
                 float mx, hipStream_t stream) {
  if (n % 8 != 0) throw std::runtime_error("clip: n % 8 != 0");
  int blocks = ew_blocks(n / 8);
  if (dtype == 0)
    hipLaunchKernelGGL((clip_kernel<_Float16>), dim3(blocks), dim3(256), 0,
                       stream, (const _Float16*)in, (_Float16*)out, n / 8,
                       mn, mx);
  else
    hipLaunchKernelGGL((clip_kernel<__bf16>), dim3(blocks), dim3(256), 0,
                       stream, (const __bf16*)in, (__bf16*)out, n / 8, mn,
                       mx);
}

// ---- tiled 2-D transpose: out[N][M] = in[M][N]^T ----
// 64x64 tiles through LDS with a +1-element row skew (65*2 B rows) so both
// the coalesced row reads and the transposed column reads are conflict-
// free. SURVEY.md §2.8 item 9 (general layout/transpose kernel).
template <typename T>
__global__ __launch_bounds__(256) void transpose2d_kernel(
    const T* __restrict__ in, T* __restrict__ out, int M, int N,
    int tiles_n) {
  __shared__ T tile[64][65];
  int tm = blockIdx.x / tiles_n, tn = blockIdx.x % tiles_n;
  int r0 = tm * 64, c0 = tn * 64;
  // 256 threads: each loads 16 rows x 4-elem chunks? simpler: 64x4 layout
  int lr = threadIdx.x & 63;        // column within the tile row group
  int lw = threadIdx.x >> 6;        // 0..3
#pragma unroll
  for (int rr = 0; rr < 16; ++rr) {
    int row = r0 + lw * 16 + rr;
    int col = c0 + lr;
    if (row < M && col < N) tile[lw * 16 + rr][lr] = in[(int64_t)row * N + col];
  }
  __syncthreads();
#pragma unroll
  for (int rr = 0; rr < 16; ++rr) {
    int orow = c0 + lw * 16 + rr;   // output row = input col
    int ocol = r0 + lr;             // output col = input row
    if (orow < N && ocol < M)
      out[(int64_t)orow * M + ocol] = tile[lr][lw * 16 + rr];
  }
}

void launch_transpose2d(int dtype, const void* in, void* out, int M, int N,
                        hipStream_t stream) {
  int tiles_m = (int)cdiv(M, 64), tiles_n = (int)cdiv(N, 64);
  dim3 grid((unsigned)(tiles_m * tiles_n));
  if (dtype == 0)
    hipLaunchKernelGGL((transpose2d_kernel<_Float16>), grid, dim3(256), 0,
                       stream, (const _Float16*)in, (_Float16*)out, M, N,
                       tiles_n);
  else
    hipLaunchKernelGGL((transpose2d_kernel<__bf16>), grid, dim3(256), 0,
                       stream, (const __bf16*)in, (__bf16*)out, M, N,
                       tiles_n);
}

// ---- strided 2-D copy: dst[m][coff + c] = src[m][c] (concat lowering) ----
// Each ONNX Concat input becomes one copy of its [M, C] block into the
// output's column range at offset coff with destination row stride ldd.
template <typename T>
__global__ void copy2d_kernel(const T* __restrict__ src, T* __restrict__ dst,
                              int64_t M, int C, int ldd) {
  int64_t total = M * (int64_t)C;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t m = i / C;
    int c = (int)(i - m * C);
    dst[m * ldd + c] = src[i];
  }
}

void launch_copy2d(int dtype, const void* src, void* dst, int64_t M, int C,
                   int ldd, int coff, hipStream_t stream) {
  int64_t total = M * (int64_t)C;
  int blocks = ew_blocks(total);
  if (dtype == 0)
    hipLaunchKernelGGL((copy2d_kernel<_Float16>), dim3(blocks), dim3(256), 0,
                       stream, (const _Float16*)src, (_Float16*)dst + coff,
                       M, C, ldd);
  else
    hipLaunchKernelGGL((copy2d_kernel<__bf16>), dim3(blocks), dim3(256), 0,
                       stream, (const __bf16*)src, (__bf16*)dst + coff, M, C,
                       ldd);
}


// ---- SwiGLU gate: out = silu(a) * b (LLaMA FFN) ----
template <typename T>
__global__ void silu_mul_kernel(const T* __restrict__ a,
                                const T* __restrict__ b,
                                T* __restrict__ out, int64_t n8) {
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n8;
       i += (int64_t)gridDim.x * blockDim.x) {
    short4v a0 = *(const short4v*)((const T*)a + i * 8);
    short4v a1 = *(const short4v*)((const T*)a + i * 8 + 4);
    short4v b0 = *(const short4v*)((const T*)b + i * 8);
    short4v b1 = *(const short4v*)((const T*)b + i * 8 + 4);
    T o[8];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float x0 = (float)((const T*)&a0)[j];
      float x1 = (float)((const T*)&a1)[j];
      o[j] = (T)(x0 / (1.f + __expf(-x0)) * (float)((const T*)&b0)[j]);
      o[4 + j] = (T)(x1 / (1.f + __expf(-x1)) * (float)((const T*)&b1)[j]);
    }
    *(short4v*)(out + i * 8) = *(const short4v*)&o[0];
    *(short4v*)(out + i * 8 + 4) = *(const short4v*)&o[4];
  }
}

void launch_silu_mul(int dtype, const void* a, const void* b, void* out,
                     int64_t n, hipStream_t stream) {
  if (n % 8 != 0) throw std::runtime_error("silu_mul: n % 8 != 0");
  int blocks = ew_blocks(n / 8);
  if (dtype == 0)
    hipLaunchKernelGGL((silu_mul_kernel<_Float16>), dim3(blocks), dim3(256),
                       0, stream, (const _Float16*)a, (const _Float16*)b,
                       (_Float16*)out, n / 8);
  else
    hipLaunchKernelGGL((silu_mul_kernel<__bf16>), dim3(blocks), dim3(256), 0,
                       stream, (const __bf16*)a, (const __bf16*)b,
                       (__bf16*)out, n / 8);
}

// ---- rotary position embedding (RoPE, LLaMA-style interleaved-half) ----
// In-place on the q and k column blocks of the fused qkv rows
// [M = B*S, 3*H*D]: for head h, pair (d, d + D/2), angle =
// pos * theta^(-2d/D). Row positions: pos = row % S (full-sequence
// forward; decode uses per-slot device pos via rope_pos).
template <typename T>
__global__ void rope_kernel(T* __restrict__ qkv,
                            const int* __restrict__ pos_dev, int M, int S,
                            int H, int D, float theta, int K /*B for dev*/) {
  int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  int half = D / 2;
  int64_t per_row = (int64_t)2 * H * half;  // q and k rotate, v untouched
  if (idx >= (int64_t)M * per_row) return;
  int64_t row = idx / per_row;
  int r = (int)(idx - row * per_row);
  int qk = r / (H * half);        // 0 = q block, 1 = k block
  int rr = r % (H * half);
  int h = rr / half, d = rr % half;
  int p;
  if (!pos_dev)
    p = (int)(row % S);                       // full-sequence forward
  else if (K > 0)
    p = pos_dev[row / K] + (int)(row % K);    // verify chunk: K rows/slot
  else
    p = pos_dev[row];                         // decode step: one row/slot
  if (p < 0 || (K > 0 && pos_dev[row / K] < 0))
    return;  // idle slot (decode)
  int hid = H * D;
  T* base = qkv + row * (int64_t)3 * hid + qk * hid + h * D;
  float ang = (float)p * __powf(theta, -2.0f * (float)d / (float)D);
  float c, sn;
  __sincosf(ang, &sn, &c);
  float x0 = (float)base[d], x1 = (float)base[d + half];
  base[d] = (T)(x0 * c - x1 * sn);
  base[d + half] = (T)(x0 * sn + x1 * c);
}

void launch_rope(int dtype, void* qkv, const void* pos_dev, int M, int S,
                 int H, int D, float theta, hipStream_t stream, int chunk) {
  if (D % 2 != 0) throw std::runtime_error("rope: D % 2 != 0");
  int64_t total = (int64_t)M * 2 * H * (D / 2);
  int blocks = (int)std::min<int64_t>((total + 255) / 256, 16384);
  if (dtype != 0) throw std::runtime_error("rope: fp16 only");
  hipLaunchKernelGGL((rope_kernel<_Float16>), dim3(blocks), dim3(256), 0,
                     stream, (_Float16*)qkv, (const int*)pos_dev, M, S, H, D,
                     theta, chunk);
}

// ---- row-wise argmax: fp16 logits [M, V] -> int32 index [M] ----
// One BLOCK (4 waves) per row, 16-B vector loads (greedy decoding head:
// 4 B/row D2H instead of the whole logits matrix; at decode M is tiny,
// so per-row parallelism matters more than row count). Ties resolve to
// the LOWEST index, matching numpy argmax so the speculative-decode
// invariance holds exactly.
__global__ __launch_bounds__(256) void argmax_rows_kernel(
    const _Float16* __restrict__ x, int* __restrict__ out, int M, int V) {
  int row = blockIdx.x;
  if (row >= M) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const _Float16* r = x + (int64_t)row * V;
  float best = -1e30f;
  int bi = 0x7fffffff;
  const int nv8 = V >> 3;
  for (int i = tid; i < nv8; i += 256) {
    short4v v0 = *(const short4v*)(r + i * 8);
    short4v v1 = *(const short4v*)(r + i * 8 + 4);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = (float)((const _Float16*)(j < 4 ? (const void*)&v0
                                                : (const void*)&v1))[j & 3];
      int c = i * 8 + j;
      if (v > best || (v == best && c < bi)) {
        best = v;
        bi = c;
      }
    }
  }
  for (int c = (nv8 << 3) + tid; c < V; c += 256) {
    float v = (float)r[c];
    if (v > best || (v == best && c < bi)) {
      best = v;
      bi = c;
    }
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) {
    float ob = __shfl_xor(best, off, 64);
    int oi = __shfl_xor(bi, off, 64);
    if (ob > best || (ob == best && oi < bi)) {
      best = ob;
      bi = oi;
    }
  }
  __shared__ float wb[4];
  __shared__ int wi[4];
  if (lane == 0) {
    wb[wave] = best;
    wi[wave] = bi;
  }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < 4; ++w)
      if (wb[w] > wb[0] || (wb[w] == wb[0] && wi[w] < wi[0])) {
        wb[0] = wb[w];
        wi[0] = wi[w];
      }
    out[row] = wi[0];
  }
}

void launch_argmax_rows(const void* x, void* out, int M, int V,
                        hipStream_t stream) {
  hipLaunchKernelGGL(argmax_rows_kernel, dim3((unsigned)M), dim3(256), 0,
                     stream, (const _Float16*)x, (int*)out, M, V);
}

// ---- device-side categorical sampling via the Gumbel-max trick ----
// argmax(logits/T + G_c) with G_c = -log(-log(u_c)) samples EXACTLY from
// softmax(logits/T) — so temperature sampling reuses the argmax shape
// and only row winners (4 B each) leave the GPU. u_c comes from a
// counter-based splitmix64 keyed on (seed[row], pos[row], c): pos is the
// DEVICE position counter, so every decode step draws fresh noise even
// inside a captured graph. temps[row] <= 0 degrades to plain greedy
// argmax (one kernel serves mixed greedy/sampled batches).
__device__ __forceinline__ uint64_t splitmix64(uint64_t z) {
  z += 0x9e3779b97f4a7c15ull;
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
  return z ^ (z >> 31);
}

__global__ __launch_bounds__(256) void gumbel_argmax_rows_kernel(
    const _Float16* __restrict__ x, int* __restrict__ out,
    const float* __restrict__ temps, const int* __restrict__ seeds,
    const int* __restrict__ pos, int M, int V) {
  int row = blockIdx.x;
  if (row >= M) return;
  const int tid = threadIdx.x;
  const int lane = tid & 63, wave = tid >> 6;
  const _Float16* r = x + (int64_t)row * V;
  const float T = temps ? temps[row] : 0.0f;
  const float invT = (T > 0.0f) ? 1.0f / T : 0.0f;
  const uint64_t key =
      ((uint64_t)(uint32_t)(seeds ? seeds[row] : 0) << 32) ^
      (uint64_t)(uint32_t)(pos ? pos[row] : 0) ^
      ((uint64_t)(uint32_t)row << 20);
  float best = -1e30f;
  int bi = 0x7fffffff;
  for (int c = tid; c < V; c += 256) {
    float v = (float)r[c];
    if (T > 0.0f) {
      uint64_t h = splitmix64(key ^ (uint64_t)c);
      // u in (0, 1): top 24 bits, never exactly 0
      float u = ((float)(uint32_t)(h >> 40) + 0.5f) * (1.0f / 16777216.f);
      v = v * invT - __logf(-__logf(u));
    }
    if (v > best || (v == best && c < bi)) {
      best = v;
      bi = c;
    }
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) {
    float ob = __shfl_xor(best, off, 64);
    int oi = __shfl_xor(bi, off, 64);
    if (ob > best || (ob == best && oi < bi)) {
      best = ob;
      bi = oi;
    }
  }
  __shared__ float wb[4];
  __shared__ int wi[4];
  if (lane == 0) {
    wb[wave] = best;
    wi[wave] = bi;
  }
  __syncthreads();
  if (tid == 0) {
    for (int w = 1; w < 4; ++w)
      if (wb[w] > wb[0] || (wb[w] == wb[0] && wi[w] < wi[0])) {
        wb[0] = wb[w];
        wi[0] = wi[w];
      }
    out[row] = wi[0];
  }
}

void launch_gumbel_argmax_rows(const void* x, void* out, const void* temps,
                               const void* seeds, const void* pos, int M,
                               int V, hipStream_t stream) {
  hipLaunchKernelGGL(gumbel_argmax_rows_kernel, dim3((unsigned)M),
                     dim3(256), 0, stream, (const _Float16*)x, (int*)out,
                     (const float*)temps, (const int*)seeds,
                     (const int*)pos, M, V);
}

}  // namespace trtlab

