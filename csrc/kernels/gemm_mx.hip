// trtlab_amd — MX-format GEMM on the CDNA4 scaled MFMA
// (mfma_scale_f32_16x16x128_f8f6f4): OCP Microscaling fp8-e4m3 elements
// with one e8m0 shared scale per 32-element K-block. This instruction is
// gfx950's headline-throughput path (K=128 per issue, 4x the K-depth of
// the 16x16x32 fp8 MFMA) and has no equivalent in the CUDA reference.
//
// C[M,N] = sum_k (a[m,k] * 2^(sa[m,k/32]-127)) * (b[n,k] * 2^(sb[n,k/32]-127))
//
// B is prepacked transposed [N][K] like every other "bt" GEMM here; scales
// are row-major u8 [M][K/32] / [N][K/32]. Operand layout (pinned down
// empirically — tools/probe_mx.py): lane-group g's 32-byte register holds
// logical k [g*16, g*16+16) (low half) and [64+g*16, 64+g*16+16) (high
// half); lane-group g's scale byte (opsel 0) applies to logical MX block
// g, i.e. k [g*32, g*32+32). Rows stage through the shared 128-byte-row
// XOR-swizzled LDS tiles (one LDS row == one instruction's K). cbsz/blgp
// select the element format (0 = fp8 e4m3; fp6/fp4 reuse the same
// instruction — planned).
#include "gemm_common.h"

namespace trtlab {

typedef __attribute__((ext_vector_type(8))) int i32x8v;

__device__ __forceinline__ void glds4(const void* gsrc, uint32_t lds_byte) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) uint32_t*)gsrc,
      (__attribute__((address_space(3))) uint32_t*)(uintptr_t)lds_byte, 4, 0,
      0);
}

// Read a 32-byte fragment (two swizzled 16-B LDS chunks) for tile row
// `row`. Lane-group g's register covers logical k [g*16, g*16+16) in its
// low 16 bytes and [64+g*16, ...) in its high 16 bytes (pinned down by
// tools/probe_mx.py: experiment E1 = 48, the split-half signature); the
// scale byte of lane-group g applies to logical block g (k [g*32,+32)).
__device__ __forceinline__ i32x8v read_frag32(const char* lds, int row,
                                              int lane) {
  uint32_t g16 = (uint32_t)(lane >> 4) * 16;
  auto swz = [&](uint32_t cb) {
    return (uint32_t)row * 128 + (cb ^ (((uint32_t)row & 7) << 4));
  };
  i32x4 lo = *(const i32x4*)(lds + swz(g16));
  i32x4 hi = *(const i32x4*)(lds + swz(64 + g16));
  i32x8v f;
  f[0] = lo[0]; f[1] = lo[1]; f[2] = lo[2]; f[3] = lo[3];
  f[4] = hi[0]; f[5] = hi[1]; f[6] = hi[2]; f[7] = hi[3];
  return f;
}

// BM = BN = 128, one workgroup = 4 waves; wave (wr, wc) = (w>>1, w&1)
// computes the [wr*64, +64) x [wc*64, +64) quadrant (4x4 fragment pairs,
// 16 scaled MFMAs per K-tile from 8 32-byte fragment reads). Double
// buffered: 2 x (128+128) rows x 128 B = 64 KiB LDS -> 2 blocks/CU.
template <typename OT, Epi E>
__global__ __launch_bounds__(256) void gemm_mxfp8_kernel(
    const uint8_t* __restrict__ A, const uint8_t* __restrict__ B,
    const uint8_t* __restrict__ Sa, const uint8_t* __restrict__ Sb,
    OT* __restrict__ C, const float* __restrict__ scale,
    const float* __restrict__ bias, int M, int N, int K, int tiles_n) {
  // 3 slots x (A tile + B tile + scale slabs): counted-vmcnt pipeline
  // keeps 2 tiles' staging in flight behind the MFMAs (9 glds per thread
  // per tile: 8 data + 1 scale u32).
  constexpr int kABytes = 128 * 128;
  constexpr int kSlot = 2 * kABytes + 1024;  // + sa[128] u32 + sb[128] u32
  __shared__ __attribute__((aligned(16))) char smem[3 * kSlot];

  uint32_t bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int m0 = (int)(bid / tiles_n) * 128;
  int n0 = (int)(bid % tiles_n) * 128;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int ktiles = K >> 7;       // 128 elems per tile
  const int kblocks = K >> 5;      // 32-elem MX blocks per row

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int f = 0; f < 4; ++f) acc[i][f] = {0.f, 0.f, 0.f, 0.f};

  auto stage = [&](int t, int slot) {
    char* base = smem + slot * kSlot;
    // A rows [m0, m0+128), B rows [n0, n0+128): 128 rows x 8 chunks each.
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int idx = c * 256 + tid;  // 0..1023
      uint32_t row = (uint32_t)idx >> 3;
      uint32_t cb = ((uint32_t)idx & 7) * 16;
      uint32_t kb = cb ^ ((row & 7) << 4);  // source-side swizzle (rule 21)
      int ar = m0 + (int)row;
      if (ar >= M) ar = M - 1;
      glds16((const char*)(A + (int64_t)ar * K + t * 128 + kb),
             (uint32_t)(uintptr_t)base + row * 128 + cb);
      int br = n0 + (int)row;
      if (br >= N) br = N - 1;
      glds16((const char*)(B + (int64_t)br * K + t * 128 + kb),
             (uint32_t)(uintptr_t)(base + kABytes) + row * 128 + cb);
    }
    // scales: one u32 (4 block-scales) per row per tile, via glds so the
    // per-thread glds count stays uniform (scattered 1-byte global loads
    // per-MFMA were the v1 bottleneck: 402 -> 747 TF when staged)
    if (tid < 128) {
      int ar = m0 + tid;
      if (ar >= M) ar = M - 1;
      glds4(Sa + (int64_t)ar * kblocks + t * 4,
            (uint32_t)(uintptr_t)(base + 2 * kABytes) + (uint32_t)tid * 4);
    } else {
      int br = n0 + tid - 128;
      if (br >= N) br = N - 1;
      glds4(Sb + (int64_t)br * kblocks + t * 4,
            (uint32_t)(uintptr_t)(base + 2 * kABytes) +
                (uint32_t)tid * 4);  // tid-128 -> second 512-B half
    }
  };

  stage(0, 0);
  if (ktiles > 1) stage(1, 1);
  for (int t = 0; t < ktiles; ++t) {
    int ahead = ktiles - 1 - t;
    if (ahead > 1) ahead = 1;
    wait_tiles_inflight<9>(ahead);
    __builtin_amdgcn_s_barrier();
    if (t + 2 < ktiles) stage(t + 2, (t + 2) % 3);
    const char* As = smem + (t % 3) * kSlot;
    const char* Bs = As + kABytes;
    const uint8_t* sas = (const uint8_t*)(As + 2 * kABytes);
    const uint8_t* sbs = sas + 512;
    int g = lane >> 4;  // lane's MX block within the tile (scale byte idx)
    i32x8v af[4];
    int sa[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int arow = wr * 64 + i * 16 + (lane & 15);
      af[i] = read_frag32(As, arow, lane);
      sa[i] = sas[arow * 4 + g];
    }
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int brow = wc * 64 + f * 16 + (lane & 15);
      i32x8v bf = read_frag32(Bs, brow, lane);
      int sb = sbs[brow * 4 + g];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        acc[i][f] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            af[i], bf, acc[i][f], 0 /*cbsz*/, 0 /*blgp*/, 0, sa[i], 0, sb);
    }
  }

#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int col = n0 + wc * 64 + f * 16 + (lane & 15);
      if (col >= N) continue;
      float sc = 1.0f, bi = 0.0f;
      if constexpr (E == Epi::kScaleBias || E == Epi::kScaleBiasRelu ||
                    E == Epi::kScaleBiasAddRelu || E == Epi::kScaleBiasGelu)
        sc = scale[col];
      if constexpr (E != Epi::kNone) bi = bias[col];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + i * 16 + ((lane >> 4) << 2) + r;
        if (row >= M) continue;
        C[(int64_t)row * N + col] =
            store_cast<OT>(apply_epi<E>(acc[i][f][r], sc, bi, 0.0f));
      }
    }
}

// Stage 64 rows through the swizzled LDS tile exactly like the GEMM
// kernel, then dump wave-0's assembled fragments so the host can verify
// the LDS round trip byte-for-byte.
__global__ __launch_bounds__(256) void mx_frag_dump_kernel(
    const uint8_t* __restrict__ A, uint8_t* __restrict__ out, int K) {
  __shared__ __attribute__((aligned(16))) char smem[64 * 128];
  const int tid = threadIdx.x;
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    int idx = c * 256 + tid;
    uint32_t row = (uint32_t)idx >> 3;
    uint32_t cb = ((uint32_t)idx & 7) * 16;
    glds16((const char*)(A + (int64_t)row * K + (cb ^ ((row & 7) << 4))),
           (uint32_t)(uintptr_t)smem + row * 128 + cb);
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  int lane = tid & 63, wave = tid >> 6;
  if (wave >= 4) return;
  i32x8v f = read_frag32(smem, wave * 16 + (lane & 15), lane);
  *(i32x8v*)(out + (wave * 64 + lane) * 32) = f;
}

void launch_mx_frag_dump(const void* A, void* out, int K, hipStream_t s) {
  hipLaunchKernelGGL(mx_frag_dump_kernel, dim3(1), dim3(256), 0, s,
                     (const uint8_t*)A, (uint8_t*)out, K);
}

// Layout probe: one wave, direct global loads, no LDS. M=N=16, K=128.
// Lane l loads its register halves from logical k [g*16,+16) and
// [64+g*16,+16); scales Sa[row][g]. Dumps acc for host-side hypothesis
// falsification (tools/probe_mx.py).
__global__ __launch_bounds__(64) void mx_probe_kernel(
    const uint8_t* __restrict__ A, const uint8_t* __restrict__ B,
    const uint8_t* __restrict__ Sa, const uint8_t* __restrict__ Sb,
    float* __restrict__ D) {
  int lane = threadIdx.x;
  int row = lane & 15, g = lane >> 4;
  i32x8v af, bf;
  {
    i32x4 lo = *(const i32x4*)(A + row * 128 + g * 16);
    i32x4 hi = *(const i32x4*)(A + row * 128 + 64 + g * 16);
    af[0] = lo[0]; af[1] = lo[1]; af[2] = lo[2]; af[3] = lo[3];
    af[4] = hi[0]; af[5] = hi[1]; af[6] = hi[2]; af[7] = hi[3];
    lo = *(const i32x4*)(B + row * 128 + g * 16);
    hi = *(const i32x4*)(B + row * 128 + 64 + g * 16);
    bf[0] = lo[0]; bf[1] = lo[1]; bf[2] = lo[2]; bf[3] = lo[3];
    bf[4] = hi[0]; bf[5] = hi[1]; bf[6] = hi[2]; bf[7] = hi[3];
  }
  int sa = Sa[row * 4 + g];
  int sb = Sb[row * 4 + g];
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(af, bf, acc, 0, 0,
                                                         0, sa, 0, sb);
#pragma unroll
  for (int r = 0; r < 4; ++r) D[((g << 2) + r) * 16 + row] = acc[r];
}

// Row-wise dynamic MXFP4 quantization of fp16 activations:
// x [M, K] fp16 -> codes [M, K/2] (low nibble = even elem) + e8m0 scales
// [M, K/32]. One thread per 32-element block (grid-strided): per-block
// amax -> shared exponent 2^(floor(log2(amax)) - 2) -> round-to-nearest
// onto the e2m1 grid {0, .5, 1, 1.5, 2, 3, 4, 6}.
__global__ __launch_bounds__(256) void quantize_mxfp4_kernel(
    const _Float16* __restrict__ x, uint8_t* __restrict__ codes,
    uint8_t* __restrict__ scales, int64_t nblocks) {
  int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= nblocks) return;
  const _Float16* src = x + b * 32;
  float v[32];
  float amax = 0.f;
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    half8v h = *(const half8v*)(src + c * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)((const _Float16*)&h)[j];
      v[c * 8 + j] = f;
      amax = fmaxf(amax, fabsf(f));
    }
  }
  int e = (amax > 0.f) ? (int)floorf(log2f(amax)) - 2 : 0;
  e = e < -127 ? -127 : (e > 127 ? 127 : e);
  scales[b] = (uint8_t)(e + 127);
  float inv = exp2f((float)-e);
  uint8_t out[16];
#pragma unroll
  for (int j = 0; j < 16; ++j) {
    uint8_t byte = 0;
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      float q = v[2 * j + h] * inv;
      float a = fabsf(q);
      // round-to-nearest onto {0,.5,1,1.5,2,3,4,6}
      int m;
      if (a < 0.25f) m = 0;
      else if (a < 0.75f) m = 1;
      else if (a < 1.25f) m = 2;
      else if (a < 1.75f) m = 3;
      else if (a < 2.5f) m = 4;
      else if (a < 3.5f) m = 5;
      else if (a < 5.0f) m = 6;
      else m = 7;
      uint8_t code = (uint8_t)(q < 0.f ? (m | 8) : m);
      byte |= (uint8_t)(code << (4 * h));
    }
    out[j] = byte;
  }
  *(f32x4*)(codes + b * 16) = *(const f32x4*)out;
}

void launch_quantize_mxfp4(const void* x, void* codes, void* scales,
                           int64_t m, int64_t k, hipStream_t stream) {
  if (k % 32 != 0)
    throw std::runtime_error("quantize_mxfp4: K must be a multiple of 32");
  int64_t nblocks = m * (k / 32);
  int64_t blocks = (nblocks + 255) / 256;
  hipLaunchKernelGGL(quantize_mxfp4_kernel, dim3((unsigned)blocks), dim3(256),
                     0, stream, (const _Float16*)x, (uint8_t*)codes,
                     (uint8_t*)scales, nblocks);
}

// Row-wise dynamic MXFP8 quantization: fp16 [M, K] -> e4m3 codes [M, K]
// + e8m0 scales [M, K/32]. One thread per 32-element block.
__global__ __launch_bounds__(256) void quantize_mxfp8_kernel(
    const _Float16* __restrict__ x, uint8_t* __restrict__ codes,
    uint8_t* __restrict__ scales, int64_t nblocks) {
  int64_t b = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (b >= nblocks) return;
  const _Float16* src = x + b * 32;
  float v[32];
  float amax = 0.f;
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    half8v h = *(const half8v*)(src + c * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = (float)((const _Float16*)&h)[j];
      v[c * 8 + j] = f;
      amax = fmaxf(amax, fabsf(f));
    }
  }
  int e = (amax > 0.f) ? (int)floorf(log2f(amax)) - 8 : 0;  // e4m3 emax 8
  e = e < -127 ? -127 : (e > 127 ? 127 : e);
  scales[b] = (uint8_t)(e + 127);
  float inv = exp2f((float)-e);
  uint8_t out[32];
#pragma unroll
  for (int j = 0; j < 32; ++j) {
    __hip_fp8_e4m3 q = store_cast<__hip_fp8_e4m3>(v[j] * inv);
    out[j] = *(const uint8_t*)&q;
  }
  *(f32x4*)(codes + b * 32) = *(const f32x4*)out;
  *(f32x4*)(codes + b * 32 + 16) = *(const f32x4*)(out + 16);
}

void launch_quantize_mxfp8(const void* x, void* codes, void* scales,
                           int64_t m, int64_t k, hipStream_t stream) {
  if (k % 32 != 0)
    throw std::runtime_error("quantize_mxfp8: K must be a multiple of 32");
  int64_t nblocks = m * (k / 32);
  int64_t blocks = (nblocks + 255) / 256;
  hipLaunchKernelGGL(quantize_mxfp8_kernel, dim3((unsigned)blocks), dim3(256),
                     0, stream, (const _Float16*)x, (uint8_t*)codes,
                     (uint8_t*)scales, nblocks);
}

// ---- MXFP4 (fp4 e2m1, e8m0 scales per 32 elems) ----
// Same scaled MFMA with cbsz/blgp = 4. fp4 operand layout is SIMPLER than
// fp8 (tools/probe_mx4.py): lane-group g's 16 packed bytes (low 4 dwords,
// upper 4 zero) cover logical k [g*32, g*32+32) contiguously, and its
// scale byte covers exactly that block. One 128-byte LDS row holds TWO
// 128-k instruction windows (256 logical k); K must be % 256.
template <typename OT, Epi E>
__global__ __launch_bounds__(256) void gemm_mxfp4_kernel(
    const uint8_t* __restrict__ A, const uint8_t* __restrict__ B,
    const uint8_t* __restrict__ Sa, const uint8_t* __restrict__ Sb,
    OT* __restrict__ C, const float* __restrict__ scale,
    const float* __restrict__ bias, int M, int N, int K, int tiles_n) {
  constexpr int kABytes = 128 * 128;
  constexpr int kSlot = 2 * kABytes + 2048;  // + 8 scale bytes/row (A, B)
  __shared__ __attribute__((aligned(16))) char smem[3 * kSlot];

  uint32_t bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int m0 = (int)(bid / tiles_n) * 128;
  int n0 = (int)(bid % tiles_n) * 128;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int kb2 = K >> 1;          // packed bytes per row
  const int ktiles = K >> 8;       // 256 logical elems per LDS tile
  const int kblocks = K >> 5;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int f = 0; f < 4; ++f) acc[i][f] = {0.f, 0.f, 0.f, 0.f};

  auto stage = [&](int t, int slot) {
    char* base = smem + slot * kSlot;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int idx = c * 256 + tid;  // 0..1023
      uint32_t row = (uint32_t)idx >> 3;
      uint32_t cb = ((uint32_t)idx & 7) * 16;
      uint32_t kb = cb ^ ((row & 7) << 4);  // source-side swizzle
      int ar = m0 + (int)row;
      if (ar >= M) ar = M - 1;
      glds16((const char*)(A + (int64_t)ar * kb2 + t * 128 + kb),
             (uint32_t)(uintptr_t)base + row * 128 + cb);
      int br = n0 + (int)row;
      if (br >= N) br = N - 1;
      glds16((const char*)(B + (int64_t)br * kb2 + t * 128 + kb),
             (uint32_t)(uintptr_t)(base + kABytes) + row * 128 + cb);
    }
    // 8 e8m0 bytes per row per 256-k tile, as four 512-B slabs
    // (A-window0 | A-window1 | B-window0 | B-window1). global_load_lds
    // packs lanes at consecutive 4-B slots from a uniform base, so each
    // slab gets its own call with a lane-natural destination.
    uint32_t sbase = (uint32_t)(uintptr_t)(base + 2 * kABytes);
    if (tid < 128) {
      int ar = m0 + tid;
      if (ar >= M) ar = M - 1;
      glds4(Sa + (int64_t)ar * kblocks + t * 8, sbase + (uint32_t)tid * 4);
      glds4(Sa + (int64_t)ar * kblocks + t * 8 + 4,
            sbase + 512 + (uint32_t)tid * 4);
    } else {
      int br = n0 + tid - 128;
      if (br >= N) br = N - 1;
      glds4(Sb + (int64_t)br * kblocks + t * 8,
            sbase + 1024 + (uint32_t)(tid - 128) * 4);
      glds4(Sb + (int64_t)br * kblocks + t * 8 + 4,
            sbase + 1536 + (uint32_t)(tid - 128) * 4);
    }
  };

  auto frag16 = [&](const char* lds, int row, int w) {
    uint32_t cb = (uint32_t)w * 64 + (uint32_t)(lane >> 4) * 16;
    i32x4 v = *(const i32x4*)(
        lds + (uint32_t)row * 128 + (cb ^ (((uint32_t)row & 7) << 4)));
    i32x8v f = {v[0], v[1], v[2], v[3], 0, 0, 0, 0};
    return f;
  };

  stage(0, 0);
  if (ktiles > 1) stage(1, 1);
  for (int t = 0; t < ktiles; ++t) {
    int ahead = ktiles - 1 - t;
    if (ahead > 1) ahead = 1;
    wait_tiles_inflight<10>(ahead);
    __builtin_amdgcn_s_barrier();
    if (t + 2 < ktiles) stage(t + 2, (t + 2) % 3);
    const char* As = smem + (t % 3) * kSlot;
    const char* Bs = As + kABytes;
    const uint8_t* sas = (const uint8_t*)(As + 2 * kABytes);
    const uint8_t* sbs = sas + 1024;
    int g = lane >> 4;
#pragma unroll
    for (int w = 0; w < 2; ++w) {
      i32x8v af[4];
      int sa[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int arow = wr * 64 + i * 16 + (lane & 15);
        af[i] = frag16(As, arow, w);
        sa[i] = sas[w * 512 + arow * 4 + g];
      }
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        int brow = wc * 64 + f * 16 + (lane & 15);
        i32x8v bf = frag16(Bs, brow, w);
        int sb = sbs[w * 512 + brow * 4 + g];
#pragma unroll
        for (int i = 0; i < 4; ++i)
          acc[i][f] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
              af[i], bf, acc[i][f], 4 /*cbsz*/, 4 /*blgp*/, 0, sa[i], 0, sb);
      }
    }
  }

#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int col = n0 + wc * 64 + f * 16 + (lane & 15);
      if (col >= N) continue;
      float sc = 1.0f, bi = 0.0f;
      if constexpr (E == Epi::kScaleBias || E == Epi::kScaleBiasRelu ||
                    E == Epi::kScaleBiasAddRelu || E == Epi::kScaleBiasGelu)
        sc = scale[col];
      if constexpr (E != Epi::kNone) bi = bias[col];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = m0 + wr * 64 + i * 16 + ((lane >> 4) << 2) + r;
        if (row >= M) continue;
        C[(int64_t)row * N + col] =
            store_cast<OT>(apply_epi<E>(acc[i][f][r], sc, bi, 0.0f));
      }
    }
}


// 64x64 small-tile MXFP4 variant (grid-starved shapes; cf. the fp8 one).
template <typename OT, Epi E>
__global__ __launch_bounds__(256) void gemm_mxfp4_small_kernel(
    const uint8_t* __restrict__ A, const uint8_t* __restrict__ B,
    const uint8_t* __restrict__ Sa, const uint8_t* __restrict__ Sb,
    OT* __restrict__ C, const float* __restrict__ scale,
    const float* __restrict__ bias, int M, int N, int K, int tiles_n) {
  constexpr int kABytes = 64 * 128;  // 64 rows x 256 logical k (packed)
  __shared__ __attribute__((aligned(16))) char smem[2 * 2 * kABytes];
  __shared__ __attribute__((aligned(4))) uint32_t sa_s[2][128], sb_s[2][128];

  uint32_t bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int m0 = (int)(bid / tiles_n) * 64;
  int n0 = (int)(bid % tiles_n) * 64;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int kb2 = K >> 1;
  const int ktiles = K >> 8;  // 256 logical elems per LDS tile
  const int kblocks = K >> 5;

  f32x4 acc[4];
#pragma unroll
  for (int f = 0; f < 4; ++f) acc[f] = {0.f, 0.f, 0.f, 0.f};

  auto stage = [&](int t, int slot) {
    char* base = smem + slot * 2 * kABytes;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = c * 256 + tid;
      uint32_t row = (uint32_t)idx >> 3;
      uint32_t cb = ((uint32_t)idx & 7) * 16;
      uint32_t kb = cb ^ ((row & 7) << 4);
      int ar = m0 + (int)row;
      if (ar >= M) ar = M - 1;
      glds16((const char*)(A + (int64_t)ar * kb2 + t * 128 + kb),
             (uint32_t)(uintptr_t)base + row * 128 + cb);
      int br = n0 + (int)row;
      if (br >= N) br = N - 1;
      glds16((const char*)(B + (int64_t)br * kb2 + t * 128 + kb),
             (uint32_t)(uintptr_t)(base + kABytes) + row * 128 + cb);
    }
    // 8 scale bytes per row per 256-k tile: two 64-u32 slabs per side
    if (tid < 64) {
      int ar = m0 + tid;
      if (ar >= M) ar = M - 1;
      sa_s[slot][tid] = *(const uint32_t*)(Sa + (int64_t)ar * kblocks + t * 8);
      sa_s[slot][64 + tid] =
          *(const uint32_t*)(Sa + (int64_t)ar * kblocks + t * 8 + 4);
    } else if (tid < 128) {
      int br = n0 + tid - 64;
      if (br >= N) br = N - 1;
      sb_s[slot][tid - 64] =
          *(const uint32_t*)(Sb + (int64_t)br * kblocks + t * 8);
      sb_s[slot][tid] =
          *(const uint32_t*)(Sb + (int64_t)br * kblocks + t * 8 + 4);
    }
  };

  auto frag16 = [&](const char* lds, int row, int w) {
    uint32_t cb = (uint32_t)w * 64 + (uint32_t)(lane >> 4) * 16;
    i32x4 v = *(const i32x4*)(
        lds + (uint32_t)row * 128 + (cb ^ (((uint32_t)row & 7) << 4)));
    i32x8v f = {v[0], v[1], v[2], v[3], 0, 0, 0, 0};
    return f;
  };

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < ktiles; ++t) {
    if (t + 1 < ktiles) stage(t + 1, cur ^ 1);
    const char* As = smem + cur * 2 * kABytes;
    const char* Bs = As + kABytes;
    int g = lane >> 4;
#pragma unroll
    for (int w = 0; w < 2; ++w) {
      int arow = wave * 16 + (lane & 15);
      i32x8v af = frag16(As, arow, w);
      int sa = ((const uint8_t*)&sa_s[cur][w * 64 + arow])[g];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        int brow = f * 16 + (lane & 15);
        i32x8v bf = frag16(Bs, brow, w);
        int sb = ((const uint8_t*)&sb_s[cur][w * 64 + brow])[g];
        acc[f] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
            af, bf, acc[f], 4, 4, 0, sa, 0, sb);
      }
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int f = 0; f < 4; ++f) {
    int col = n0 + f * 16 + (lane & 15);
    if (col >= N) continue;
    float sc = 1.0f, bi = 0.0f;
    if constexpr (E == Epi::kScaleBias || E == Epi::kScaleBiasRelu ||
                  E == Epi::kScaleBiasAddRelu || E == Epi::kScaleBiasGelu)
      sc = scale[col];
    if constexpr (E != Epi::kNone) bi = bias[col];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = m0 + wave * 16 + ((lane >> 4) << 2) + r;
      if (row >= M) continue;
      C[(int64_t)row * N + col] =
          store_cast<OT>(apply_epi<E>(acc[f][r], sc, bi, 0.0f));
    }
  }
}

// out_dtype: 0 = fp16 (engine path, with epilogue), 2 = fp32 raw C.
void launch_gemm_mxfp4(const void* A, const void* B, const void* Sa,
                       const void* Sb, void* C, int M, int N, int K,
                       hipStream_t stream, int out_dtype, int epi,
                       const float* scale, const float* bias) {
  if (K % 256 != 0)
    throw std::runtime_error("gemm_mxfp4: K must be a multiple of 256");
  int tiles_n = (int)cdiv(N, 128);
  dim3 grid((unsigned)(cdiv(M, 128) * tiles_n));
  if ((long)cdiv(M, 128) * tiles_n < 64) {
    // grid-starved: 4x the workgroups at 64x64
    int tn64 = (int)cdiv(N, 64);
    dim3 g64((unsigned)(cdiv(M, 64) * tn64));
    if (out_dtype == 2) {
      hipLaunchKernelGGL((gemm_mxfp4_small_kernel<float, Epi::kNone>), g64,
                         dim3(256), 0, stream, (const uint8_t*)A,
                         (const uint8_t*)B, (const uint8_t*)Sa,
                         (const uint8_t*)Sb, (float*)C, nullptr, nullptr, M,
                         N, K, tn64);
      return;
    }
    epi_dispatch(epi, [&](auto e) {
      constexpr Epi EE = decltype(e)::value;
      hipLaunchKernelGGL((gemm_mxfp4_small_kernel<_Float16, EE>), g64,
                         dim3(256), 0, stream, (const uint8_t*)A,
                         (const uint8_t*)B, (const uint8_t*)Sa,
                         (const uint8_t*)Sb, (_Float16*)C, scale, bias, M, N,
                         K, tn64);
    });
    return;
  }
  if (out_dtype == 2) {
    hipLaunchKernelGGL((gemm_mxfp4_kernel<float, Epi::kNone>), grid,
                       dim3(256), 0, stream, (const uint8_t*)A,
                       (const uint8_t*)B, (const uint8_t*)Sa,
                       (const uint8_t*)Sb, (float*)C, nullptr, nullptr, M, N,
                       K, tiles_n);
    return;
  }
  epi_dispatch(epi, [&](auto e) {
    constexpr Epi EE = decltype(e)::value;
    hipLaunchKernelGGL((gemm_mxfp4_kernel<_Float16, EE>), grid, dim3(256), 0,
                       stream, (const uint8_t*)A, (const uint8_t*)B,
                       (const uint8_t*)Sa, (const uint8_t*)Sb, (_Float16*)C,
                       scale, bias, M, N, K, tiles_n);
  });
}

// fp4 variant of the layout probe: A/B rows are 64 packed bytes (2 elems
// per byte); lane-group g is assumed to cover logical k [g*16,+16) and
// [64+g*16,+16) like fp8 -> bytes [g*8,+8) and [32+g*8,+8), passed in the
// low 4 dwords (CK: f4 operands occupy v4i32 zero-extended to v8i32).
__global__ __launch_bounds__(64) void mx4_probe_kernel(
    const uint8_t* __restrict__ A, const uint8_t* __restrict__ B,
    const uint8_t* __restrict__ Sa, const uint8_t* __restrict__ Sb,
    float* __restrict__ D) {
  int lane = threadIdx.x;
  int row = lane & 15, g = lane >> 4;
  i32x8v af = {0, 0, 0, 0, 0, 0, 0, 0}, bf = {0, 0, 0, 0, 0, 0, 0, 0};
  {
    // contiguous mapping: lane-group g holds logical elems [g*32,+32)
    // (= 16 packed bytes) and its scale byte covers exactly that block
    i32x4 v = *(const i32x4*)(A + row * 64 + g * 16);
    af[0] = v[0]; af[1] = v[1]; af[2] = v[2]; af[3] = v[3];
    v = *(const i32x4*)(B + row * 64 + g * 16);
    bf[0] = v[0]; bf[1] = v[1]; bf[2] = v[2]; bf[3] = v[3];
  }
  int sa = Sa[row * 4 + g];
  int sb = Sb[row * 4 + g];
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(af, bf, acc, 4, 4,
                                                         0, sa, 0, sb);
#pragma unroll
  for (int r = 0; r < 4; ++r) D[((g << 2) + r) * 16 + row] = acc[r];
}

void launch_mx4_probe(const void* A, const void* B, const void* Sa,
                      const void* Sb, void* D, hipStream_t stream) {
  hipLaunchKernelGGL(mx4_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const uint8_t*)A, (const uint8_t*)B, (const uint8_t*)Sa,
                     (const uint8_t*)Sb, (float*)D);
}

void launch_mx_probe(const void* A, const void* B, const void* Sa,
                     const void* Sb, void* D, hipStream_t stream) {
  hipLaunchKernelGGL(mx_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const uint8_t*)A, (const uint8_t*)B, (const uint8_t*)Sa,
                     (const uint8_t*)Sb, (float*)D);
}


// 64x64 small-tile MXFP8 variant for grid-starved serving shapes (e.g.
// BERT b8: M=1024 gives the 128-tile kernel only 48 workgroups on 256
// CUs). 4 waves x [16 rows x 64 cols]; 2-phase staging (latency-bound
// shapes; the deep pipeline is the 128-tile's job).
template <typename OT, Epi E>
__global__ __launch_bounds__(256) void gemm_mxfp8_small_kernel(
    const uint8_t* __restrict__ A, const uint8_t* __restrict__ B,
    const uint8_t* __restrict__ Sa, const uint8_t* __restrict__ Sb,
    OT* __restrict__ C, const float* __restrict__ scale,
    const float* __restrict__ bias, int M, int N, int K, int tiles_n) {
  constexpr int kABytes = 64 * 128;
  __shared__ __attribute__((aligned(16))) char smem[2 * 2 * kABytes];
  __shared__ __attribute__((aligned(4))) uint32_t sa_s[2][64], sb_s[2][64];

  uint32_t bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int m0 = (int)(bid / tiles_n) * 64;
  int n0 = (int)(bid % tiles_n) * 64;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int ktiles = K >> 7;
  const int kblocks = K >> 5;

  f32x4 acc[4];
#pragma unroll
  for (int f = 0; f < 4; ++f) acc[f] = {0.f, 0.f, 0.f, 0.f};

  auto stage = [&](int t, int slot) {
    char* base = smem + slot * 2 * kABytes;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = c * 256 + tid;  // 0..511
      uint32_t row = (uint32_t)idx >> 3;
      uint32_t cb = ((uint32_t)idx & 7) * 16;
      uint32_t kb = cb ^ ((row & 7) << 4);
      int ar = m0 + (int)row;
      if (ar >= M) ar = M - 1;
      glds16((const char*)(A + (int64_t)ar * K + t * 128 + kb),
             (uint32_t)(uintptr_t)base + row * 128 + cb);
      int br = n0 + (int)row;
      if (br >= N) br = N - 1;
      glds16((const char*)(B + (int64_t)br * K + t * 128 + kb),
             (uint32_t)(uintptr_t)(base + kABytes) + row * 128 + cb);
    }
    if (tid < 64) {
      int ar = m0 + tid;
      if (ar >= M) ar = M - 1;
      sa_s[slot][tid] = *(const uint32_t*)(Sa + (int64_t)ar * kblocks + t * 4);
    } else if (tid < 128) {
      int br = n0 + tid - 64;
      if (br >= N) br = N - 1;
      sb_s[slot][tid - 64] =
          *(const uint32_t*)(Sb + (int64_t)br * kblocks + t * 4);
    }
  };

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < ktiles; ++t) {
    if (t + 1 < ktiles) stage(t + 1, cur ^ 1);
    const char* As = smem + cur * 2 * kABytes;
    const char* Bs = As + kABytes;
    int g = lane >> 4;
    i32x8v af = read_frag32(As, wave * 16 + (lane & 15), lane);
    int sa = ((const uint8_t*)&sa_s[cur][wave * 16 + (lane & 15)])[g];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      i32x8v bf = read_frag32(Bs, f * 16 + (lane & 15), lane);
      int sb = ((const uint8_t*)&sb_s[cur][f * 16 + (lane & 15)])[g];
      acc[f] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
          af, bf, acc[f], 0, 0, 0, sa, 0, sb);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int f = 0; f < 4; ++f) {
    int col = n0 + f * 16 + (lane & 15);
    if (col >= N) continue;
    float sc = 1.0f, bi = 0.0f;
    if constexpr (E == Epi::kScaleBias || E == Epi::kScaleBiasRelu ||
                  E == Epi::kScaleBiasAddRelu || E == Epi::kScaleBiasGelu)
      sc = scale[col];
    if constexpr (E != Epi::kNone) bi = bias[col];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = m0 + wave * 16 + ((lane >> 4) << 2) + r;
      if (row >= M) continue;
      C[(int64_t)row * N + col] =
          store_cast<OT>(apply_epi<E>(acc[f][r], sc, bi, 0.0f));
    }
  }
}

// out_dtype: 0 = fp16 (engine path, with epilogue), 2 = fp32 raw C.
void launch_gemm_mxfp8(const void* A, const void* B, const void* Sa,
                       const void* Sb, void* C, int M, int N, int K,
                       hipStream_t stream, int out_dtype, int epi,
                       const float* scale, const float* bias) {
  if (K % 128 != 0)
    throw std::runtime_error("gemm_mxfp8: K must be a multiple of 128");
  int tiles_n = (int)cdiv(N, 128);
  dim3 grid((unsigned)(cdiv(M, 128) * tiles_n));
  if ((long)cdiv(M, 128) * tiles_n < 64) {
    // grid-starved: 4x the workgroups at 64x64
    int tn64 = (int)cdiv(N, 64);
    dim3 g64((unsigned)(cdiv(M, 64) * tn64));
    if (out_dtype == 2) {
      hipLaunchKernelGGL((gemm_mxfp8_small_kernel<float, Epi::kNone>), g64,
                         dim3(256), 0, stream, (const uint8_t*)A,
                         (const uint8_t*)B, (const uint8_t*)Sa,
                         (const uint8_t*)Sb, (float*)C, nullptr, nullptr, M,
                         N, K, tn64);
      return;
    }
    epi_dispatch(epi, [&](auto e) {
      constexpr Epi EE = decltype(e)::value;
      hipLaunchKernelGGL((gemm_mxfp8_small_kernel<_Float16, EE>), g64,
                         dim3(256), 0, stream, (const uint8_t*)A,
                         (const uint8_t*)B, (const uint8_t*)Sa,
                         (const uint8_t*)Sb, (_Float16*)C, scale, bias, M, N,
                         K, tn64);
    });
    return;
  }
  if (out_dtype == 2) {
    hipLaunchKernelGGL((gemm_mxfp8_kernel<float, Epi::kNone>), grid,
                       dim3(256), 0, stream, (const uint8_t*)A,
                       (const uint8_t*)B, (const uint8_t*)Sa,
                       (const uint8_t*)Sb, (float*)C, nullptr, nullptr, M, N,
                       K, tiles_n);
    return;
  }
  epi_dispatch(epi, [&](auto e) {
    constexpr Epi EE = decltype(e)::value;
    hipLaunchKernelGGL((gemm_mxfp8_kernel<_Float16, EE>), grid, dim3(256), 0,
                       stream, (const uint8_t*)A, (const uint8_t*)B,
                       (const uint8_t*)Sa, (const uint8_t*)Sb, (_Float16*)C,
                       scale, bias, M, N, K, tiles_n);
  });
}

}  // namespace trtlab
