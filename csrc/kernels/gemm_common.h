// trtlab_amd — shared MFMA-GEMM machinery for gfx950 kernels.
// See gemm.hip for the structure notes.
#pragma once
#include <hip/hip_fp8.h>

#include "../common.h"

namespace trtlab {

// ---------------------------------------------------------------- epilogues
enum class Epi : int {
  kNone = 0,           // C = acc
  kBias = 1,           // C = acc + bias[n]
  kBiasRelu = 2,       // C = relu(acc + bias[n])
  kBiasGelu = 3,       // C = gelu(acc + bias[n])  (tanh approximation)
  kScaleBias = 4,      // C = acc*scale[n] + bias[n]          (folded BN)
  kScaleBiasRelu = 5,  // C = relu(acc*scale[n] + bias[n])    (conv+BN+ReLU)
  kScaleBiasAddRelu = 6,  // C = relu(acc*scale[n]+bias[n]+res[m][n])
  kScaleBiasGelu = 7,     // C = gelu(acc*scale[n] + bias[n])  (fp8 gemms)
};

__device__ __forceinline__ float gelu_tanh(float x) {
  const float k0 = 0.7978845608028654f;
  const float k1 = 0.044715f;
  float u = k0 * (x + k1 * x * x * x);
  return 0.5f * x * (1.0f + tanhf(u));
}

template <Epi E>
__device__ __forceinline__ float apply_epi(float acc, float scale, float bias,
                                           float res) {
  float v = acc;
  if constexpr (E == Epi::kScaleBias || E == Epi::kScaleBiasRelu ||
                E == Epi::kScaleBiasAddRelu || E == Epi::kScaleBiasGelu)
    v = v * scale;
  if constexpr (E != Epi::kNone) v = v + bias;
  if constexpr (E == Epi::kScaleBiasAddRelu) v = v + res;
  if constexpr (E == Epi::kBiasRelu || E == Epi::kScaleBiasRelu ||
                E == Epi::kScaleBiasAddRelu)
    v = fmaxf(v, 0.0f);
  if constexpr (E == Epi::kBiasGelu || E == Epi::kScaleBiasGelu)
    v = gelu_tanh(v);
  return v;
}

// ------------------------------------------------------------- MFMA dispatch
// All dtypes share the same 128-byte LDS tile row and 16-B fragment reads:
// fp16/bf16 use mfma_f32_16x16x32 (K=32 elems/instr, 2 per row half),
// int8 uses mfma_i32_16x16x64 (K=64 elems/instr, 2 per row half) — the
// byte-level addressing is identical, only the element width differs.
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8v;
typedef __attribute__((ext_vector_type(16))) char i8x16v;
typedef __attribute__((ext_vector_type(4))) int i32x4;
typedef __attribute__((ext_vector_type(2))) int i32x2;

template <typename T>
struct Mfma16x16x32;

// kFragBytes: bytes of one lane's A/B fragment (ds_read width);
// kStepBytes: bytes of one MFMA K-step along a 128-B LDS row.
template <>
struct Mfma16x16x32<_Float16> {
  using frag = half8v;
  using accv = f32x4;
  static constexpr int kFragBytes = 16, kStepBytes = 64;
  static __device__ __forceinline__ f32x4 run(frag a, frag b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_f16(a, b, c, 0, 0, 0);
  }
};

template <>
struct Mfma16x16x32<__bf16> {
  using frag = bf16x8v;
  using accv = f32x4;
  static constexpr int kFragBytes = 16, kStepBytes = 64;
  static __device__ __forceinline__ f32x4 run(frag a, frag b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
  }
};

template <>
struct Mfma16x16x32<int8_t> {
  using frag = i8x16v;
  using accv = i32x4;
  static constexpr int kFragBytes = 16, kStepBytes = 64;
  static __device__ __forceinline__ i32x4 run(frag a, frag b, i32x4 c) {
    return __builtin_amdgcn_mfma_i32_16x16x64_i8(a, b, c, 0, 0, 0);
  }
};

// OCP fp8 e4m3 (gfx950-native; NOT the MI300X fnuz variant). K=32 per
// instruction, 8-byte lane fragments (ds_read_b64); 4 K-steps per 128-B row.
template <>
struct Mfma16x16x32<__hip_fp8_e4m3> {
  using frag = long;
  using accv = f32x4;
  static constexpr int kFragBytes = 8, kStepBytes = 32;
  static __device__ __forceinline__ f32x4 run(frag a, frag b, f32x4 c) {
    return __builtin_amdgcn_mfma_f32_16x16x32_fp8_fp8(a, b, c, 0, 0, 0);
  }
};

// Elements per 128-byte K-tile row (the staged BK).
template <typename T>
constexpr int kTileElems = 128 / (int)sizeof(T);

// Output store conversion: plain cast for fp16/bf16; round+clamp for int8.
template <typename T>
__device__ __forceinline__ T store_cast(float v) {
  return (T)v;
}
template <>
__device__ __forceinline__ int8_t store_cast<int8_t>(float v) {
  float r = rintf(v);
  r = fminf(fmaxf(r, -127.f), 127.f);
  return (int8_t)r;
}
template <>
__device__ __forceinline__ __hip_fp8_e4m3 store_cast<__hip_fp8_e4m3>(float v) {
  return __hip_fp8_e4m3(fminf(fmaxf(v, -448.f), 448.f));
}

// ------------------------------------------------------------------ staging
__device__ __forceinline__ void glds16(const void* gsrc, uint32_t lds_byte) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) uint32_t*)gsrc,
      (__attribute__((address_space(3))) uint32_t*)(uintptr_t)lds_byte, 16, 0,
      0);
}

// Stage a ROWS x 64-elem tile of a row-major matrix into LDS, rows clamped
// to [0, nrows). Source-side XOR swizzle (rule 21). ROWS in {64, 128}.
template <typename T, int ROWS>
__device__ __forceinline__ void stage_tile(const T* __restrict__ src,
                                           int64_t stride_elems, int row0,
                                           int nrows, uint32_t lds_base,
                                           int tid) {
#pragma unroll
  for (int c = 0; c < ROWS / 32; ++c) {
    uint32_t p = c * 4096 + tid * 16;
    uint32_t row = p >> 7;
    uint32_t kb = (p & 127) ^ ((row & 7) << 4);
    int r = row0 + (int)row;
    r = (r < 0) ? 0 : (r >= nrows ? nrows - 1 : r);
    const char* g = (const char*)src +
                    ((int64_t)r - row0) * stride_elems * (int64_t)sizeof(T) +
                    kb;
    glds16(g, lds_base + p);
  }
}

// Swizzled ds_read of one MFMA fragment (8 consecutive k elements, 8 or
// 16 B per lane). The XOR swizzle flips bits 4-6 of the byte offset, so
// 8- and 16-byte-aligned fragments stay naturally aligned and contiguous.
template <typename T>
__device__ __forceinline__ typename Mfma16x16x32<T>::frag read_frag(
    const char* lds, uint32_t row, uint32_t kbyte) {
  uint32_t off = (row << 7) + (kbyte ^ ((row & 7) << 4));
  return *(const typename Mfma16x16x32<T>::frag*)(lds + off);
}

// ---------------------------------------------------------- tiled compute
// Shared MFMA core for gemm/conv: 4 waves as 2x2, wave tile (BM/2)x(BN/2),
// fragments of 16x16x32, BK=64 (two MFMA k-steps per staged tile).
template <typename T, int BM, int BN>
__device__ __forceinline__ void mfma_tile(
    const char* As, const char* Bs, int lane, int wr, int wc,
    typename Mfma16x16x32<T>::accv (&acc)[BM / 32][BN / 32]) {
  using MF = Mfma16x16x32<T>;
  constexpr int MFr = BM / 32, NFr = BN / 32;
  constexpr int kSteps = 128 / MF::kStepBytes;
#pragma unroll
  for (int ks = 0; ks < kSteps; ++ks) {
    typename MF::frag af[MFr], bf[NFr];
    uint32_t kbyte = ks * MF::kStepBytes + ((lane >> 4) * MF::kFragBytes);
#pragma unroll
    for (int f = 0; f < MFr; ++f)
      af[f] = read_frag<T>(As, wr * (BM / 2) + f * 16 + (lane & 15), kbyte);
#pragma unroll
    for (int f = 0; f < NFr; ++f)
      bf[f] = read_frag<T>(Bs, wc * (BN / 2) + f * 16 + (lane & 15), kbyte);
#pragma unroll
    for (int i = 0; i < MFr; ++i)
#pragma unroll
      for (int j = 0; j < NFr; ++j)
        acc[i][j] = MF::run(af[i], bf[j], acc[i][j]);
  }
}

// Shared predicated epilogue store. D mapping for 16x16x32 MFMA:
// col = lane&15, row = (lane>>4)*4 + r. OT = output/residual element type
// (defaults to the compute type; fp8-in/fp16-out gemms set OT=_Float16).
template <typename T, Epi E, int BM, int BN, typename OT = T>
__device__ __forceinline__ void store_epilogue(
    typename Mfma16x16x32<T>::accv (&acc)[BM / 32][BN / 32],
    OT* __restrict__ C, int64_t ldc, int m0, int n0, int M, int N,
    const float* __restrict__ scale, const float* __restrict__ bias,
    const OT* __restrict__ residual, float res_scale, int lane, int wr,
    int wc, float out_scale = 1.0f) {
  constexpr int MFr = BM / 32, NFr = BN / 32;
#pragma unroll
  for (int i = 0; i < MFr; ++i) {
#pragma unroll
    for (int j = 0; j < NFr; ++j) {
      int col = n0 + wc * (BN / 2) + j * 16 + (lane & 15);
      if (col >= N) continue;
      float sc = 1.0f, bi = 0.0f;
      if constexpr (E == Epi::kScaleBias || E == Epi::kScaleBiasRelu ||
                    E == Epi::kScaleBiasAddRelu || E == Epi::kScaleBiasGelu)
        sc = scale[col];
      if constexpr (E != Epi::kNone) bi = bias[col];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * (BM / 2) + i * 16 + ((lane >> 4) << 2) + r;
        if (row >= M) continue;
        float res = 0.0f;
        if constexpr (E == Epi::kScaleBiasAddRelu)
          res = (float)residual[(int64_t)row * ldc + col] * res_scale;
        float v = apply_epi<E>((float)acc[i][j][r], sc, bi, res);
        C[(int64_t)row * ldc + col] = store_cast<OT>(v * out_scale);
      }
    }
  }
}

// ------------------------------------------------- deep staging pipeline
// Counted-vmcnt wait: block until at most `ahead` tiles' glds remain in
// flight (G = glds instructions per thread per tile). vmcnt retires in
// issue order, so <= G*ahead outstanding means the current tile landed.
// (cdna guide §5 "Pipelining across barriers": hipcc's __syncthreads
// drains glds with vmcnt(0); the counted wait + raw s_barrier keeps
// later tiles' loads in flight across the barrier.)
template <int G>
__device__ __forceinline__ void wait_tiles_inflight(int ahead) {
  if (ahead >= 2) {
    if constexpr (G == 2)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else if constexpr (G == 3)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else if constexpr (G == 4)
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    else if constexpr (G == 6)
      asm volatile("s_waitcnt vmcnt(12)" ::: "memory");
    else if constexpr (G == 9)
      asm volatile("s_waitcnt vmcnt(18)" ::: "memory");
    else if constexpr (G == 10)
      asm volatile("s_waitcnt vmcnt(20)" ::: "memory");
    else if constexpr (G == 12)
      asm volatile("s_waitcnt vmcnt(24)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(16)" ::: "memory");
  } else if (ahead == 1) {
    if constexpr (G == 2)
      asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
    else if constexpr (G == 3)
      asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
    else if constexpr (G == 4)
      asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
    else if constexpr (G == 6)
      asm volatile("s_waitcnt vmcnt(6)" ::: "memory");
    else if constexpr (G == 9)
      asm volatile("s_waitcnt vmcnt(9)" ::: "memory");
    else if constexpr (G == 10)
      asm volatile("s_waitcnt vmcnt(10)" ::: "memory");
    else if constexpr (G == 12)
      asm volatile("s_waitcnt vmcnt(12)" ::: "memory");
    else
      asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
  } else {
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  }
}

// Pick the pipeline depth: grid-starved shapes (fewer blocks than ~1.5x
// the CU count) can't hide the per-K-tile staging latency across blocks,
// so spend 4x LDS on a 3-deep in-block pipeline instead.
inline bool want_deep_pipe(long blocks, int ktiles) {
  static const long gate = [] {
    const char* e = getenv("TRTLAB_DEEP_GATE");  // tuning experiments
    return e ? atol(e) : 384L;
  }();
  return blocks < gate && ktiles >= 2;
}

// Split-K slab store: this (tile, slice)'s [BM][BN] fp32 partial sums.
template <typename T, int BM, int BN>
__device__ __forceinline__ void store_splitk(
    typename Mfma16x16x32<T>::accv (&acc)[BM / 32][BN / 32],
    float* __restrict__ slab, int lane, int wr, int wc) {
#pragma unroll
  for (int i = 0; i < BM / 32; ++i)
#pragma unroll
    for (int j = 0; j < BN / 32; ++j) {
      int col = wc * (BN / 2) + j * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = wr * (BM / 2) + i * 16 + ((lane >> 4) << 2) + r;
        slab[row * BN + col] = (float)acc[i][j][r];
      }
    }
}

// Split-K decision: only for severely grid-starved, K-heavy shapes (the
// slab round-trip + reduce launch costs ~8-10 us, so marginal cases lose —
// measured in tools/tune_tiles.py). Returns 1 = no split.
inline int pick_splitk(long blocks, int ktiles) {
  if (blocks > 144 || ktiles < 24) return 1;
  int splitk = 1;
  while (blocks * splitk < 256 && ktiles / (splitk * 2) >= 4 && splitk < 8)
    splitk *= 2;
  return splitk;
}

// Plain GEMMs amortize the reduce pass better than convs (contiguous A,
// no im2col address work in the partials), so the gate is wider: BERT's
// K=768 projections (12 K-tiles, 48-144 blocks at M=1024) qualify.
inline int pick_splitk_gemm(long blocks, int ktiles) {
  if (blocks > 144 || ktiles < 8) return 1;
  int splitk = 1;
  while (blocks * splitk < 256 && ktiles / (splitk * 2) >= 3 && splitk < 8)
    splitk *= 2;
  return splitk;
}

// Host-side tile-config choice: prefer the config that fills the chip
// (>=512 workgroups) at the highest tile utilization; otherwise maximize
// parallelism x utilization. Small deep-layer shapes (ResNet stage 4/5 at
// batch 8) need 64x64 tiles to reach enough workgroups.
struct TileCfg { int bm; int bn; };
inline TileCfg pick_tile(int M, int N) {
  const TileCfg cands[] = {{128, 128}, {128, 64}, {64, 128}, {64, 64}};
  TileCfg best = cands[0];
  double best_score = -1.0;
  for (const auto& c : cands) {
    long tm = cdiv(M, c.bm), tn = cdiv(N, c.bn);
    double par = (double)std::min<long>(tm * tn, 512) / 512.0;
    double util = (double)M * N / ((double)tm * c.bm * tn * c.bn);
    double score = par * util;
    if (score > best_score + 1e-9) {
      best_score = score;
      best = c;
    }
  }
  return best;
}

// Explicit tile override codes: 1=(128,128) 2=(128,64) 3=(64,128)
// 4=(64,64) 5=(256,128) (GEMM-only: halves staged bytes per FLOP for the
// staging-bandwidth-bound large-M transformer shapes; conv clamps it).
inline TileCfg tile_from_code(int code) {
  switch (code) {
    case 1: return {128, 128};
    case 2: return {128, 64};
    case 3: return {64, 128};
    case 5: return {256, 128};
    default: return {64, 64};
  }
}

// Dispatch a runtime TileCfg to a compile-time <BM, BN> template call.
template <typename F>
inline void tile_dispatch(TileCfg c, F&& f) {
  if (c.bm == 128 && c.bn == 128) f(std::integral_constant<int, 128>{}, std::integral_constant<int, 128>{});
  else if (c.bm == 128) f(std::integral_constant<int, 128>{}, std::integral_constant<int, 64>{});
  else if (c.bn == 128) f(std::integral_constant<int, 64>{}, std::integral_constant<int, 128>{});
  else f(std::integral_constant<int, 64>{}, std::integral_constant<int, 64>{});
}

// Bijective XCD-aware blockIdx remap (T1).
__device__ __forceinline__ uint32_t xcd_swizzle(uint32_t bid, uint32_t nwg) {
  uint32_t q = nwg / 8, r = nwg % 8;
  uint32_t xcd = bid % 8, idx = bid / 8;
  return (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
}

// Dispatch a runtime Epi to a compile-time template instantiation.
template <typename F>
inline void epi_dispatch(int epi, F&& f) {
  switch ((Epi)epi) {
    case Epi::kNone: f(std::integral_constant<Epi, Epi::kNone>{}); break;
    case Epi::kBias: f(std::integral_constant<Epi, Epi::kBias>{}); break;
    case Epi::kBiasRelu: f(std::integral_constant<Epi, Epi::kBiasRelu>{}); break;
    case Epi::kBiasGelu: f(std::integral_constant<Epi, Epi::kBiasGelu>{}); break;
    case Epi::kScaleBias: f(std::integral_constant<Epi, Epi::kScaleBias>{}); break;
    case Epi::kScaleBiasRelu: f(std::integral_constant<Epi, Epi::kScaleBiasRelu>{}); break;
    case Epi::kScaleBiasAddRelu: f(std::integral_constant<Epi, Epi::kScaleBiasAddRelu>{}); break;
    case Epi::kScaleBiasGelu: f(std::integral_constant<Epi, Epi::kScaleBiasGelu>{}); break;
  }
}

}  // namespace trtlab
