// trtlab_amd — common HIP helpers for gfx950 (CDNA4) kernels.
// MI355X-native: wave64, MFMA, LDS-tiled. No CUDA compatibility paths.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>
#include <algorithm>
#include <cstdint>
#include <cstdio>
#include <stdexcept>
#include <string>
#include <type_traits>

#define TRT_HIP_CHECK(expr)                                                    \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess) {                                                    \
      throw std::runtime_error(std::string("HIP error: ") +                    \
                               hipGetErrorString(_e) + " at " + __FILE__ +     \
                               ":" + std::to_string(__LINE__) + " in " +       \
                               #expr);                                         \
    }                                                                          \
  } while (0)

namespace trtlab {

// Wave width on CDNA4 is 64 (not 32). Hard-coded per the gfx950 ABI.
constexpr int kWave = 64;

// ---- vector types for wide loads (G13: always vectorize fp16/bf16) ----
typedef __attribute__((ext_vector_type(4))) short short4v;
typedef __attribute__((ext_vector_type(8))) short short8v;   // 16 B/lane
typedef __attribute__((ext_vector_type(2))) float float2v;
typedef __attribute__((ext_vector_type(4))) float float4v;
typedef __attribute__((ext_vector_type(4))) unsigned int uint4v;

// MFMA fragment types (gfx950 16x16x32 f16/bf16: 8 elems in, 4 f32 out).
typedef __attribute__((ext_vector_type(8))) _Float16 half8v;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;

// Exact unsigned division by a runtime constant via multiply-shift.
// Valid for x < 2^24 and 1 <= d < 2^24 (host asserts sizes).
struct FastDiv {
  uint64_t mul;  // ceil(2^48 / d)
  uint32_t d;
};

__host__ inline FastDiv make_fastdiv(uint32_t d) {
  FastDiv f;
  f.d = d;
  f.mul = (((__uint128_t)1 << 48) + d - 1) / d;
  return f;
}

__device__ __forceinline__ uint32_t fdiv(uint32_t x, const FastDiv f) {
  return (uint32_t)(((uint64_t)x * f.mul) >> 48);
}
__device__ __forceinline__ uint32_t fmod(uint32_t x, const FastDiv f) {
  return x - fdiv(x, f) * f.d;
}

// ceil-div / round-up helpers (host+device)
__host__ __device__ inline int64_t cdiv(int64_t a, int64_t b) { return (a + b - 1) / b; }
__host__ __device__ inline int64_t round_up(int64_t a, int64_t b) { return cdiv(a, b) * b; }

}  // namespace trtlab
