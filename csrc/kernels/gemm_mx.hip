// trtlab_amd — MX-format GEMM on the CDNA4 scaled MFMA
// (mfma_scale_f32_16x16x128_f8f6f4): OCP Microscaling fp8-e4m3 elements
// with one e8m0 shared scale per 32-element K-block. This instruction is
// gfx950's headline-throughput path (K=128 per issue, 4x the K-depth of
// the 16x16x32 fp8 MFMA) and has no equivalent in the CUDA reference.
//
// C[M,N] = sum_k (a[m,k] * 2^(sa[m,k/32]-127)) * (b[n,k] * 2^(sb[n,k/32]-127))
//
// B is prepacked transposed [N][K] like every other "bt" GEMM here; scales
// are row-major u8 [M][K/32] / [N][K/32]. A/B fragments: 32 bytes per lane
// (row = lane&15, k = (lane>>4)*32 + j — the 16x16x32 layout scaled 4x in
// K), staged through the shared 128-byte-row XOR-swizzled LDS tiles
// (one LDS row == one instruction's K). Lane's 32 elements are exactly one
// MX block, so the per-lane scale operand is the block scale, selected
// via opsel byte 0. cbsz/blgp = 0 = fp8 e4m3 for both operands (fp6/fp4
// use the same instruction with different format codes — planned).
#include "gemm_common.h"

namespace trtlab {

typedef __attribute__((ext_vector_type(8))) int i32x8v;

// Read a 32-byte fragment (two swizzled 16-B LDS chunks) for tile row
// `row` at K-chunk `(lane>>4)`.
__device__ __forceinline__ i32x8v read_frag32(const char* lds, int row,
                                              int lane) {
  uint32_t cb0 = (uint32_t)(lane >> 4) * 32;
  auto swz = [&](uint32_t cb) {
    return (uint32_t)row * 128 + (cb ^ (((uint32_t)row & 7) << 4));
  };
  i32x4 lo = *(const i32x4*)(lds + swz(cb0));
  i32x4 hi = *(const i32x4*)(lds + swz(cb0 + 16));
  i32x8v f;
  f[0] = lo[0]; f[1] = lo[1]; f[2] = lo[2]; f[3] = lo[3];
  f[4] = hi[0]; f[5] = hi[1]; f[6] = hi[2]; f[7] = hi[3];
  return f;
}

// BM = BN = 64, one workgroup = 4 waves; wave w computes rows
// [w*16, w*16+16) x all 64 cols. One K-tile (128 elems) per MFMA issue.
__global__ __launch_bounds__(256) void gemm_mxfp8_kernel(
    const uint8_t* __restrict__ A, const uint8_t* __restrict__ B,
    const uint8_t* __restrict__ Sa, const uint8_t* __restrict__ Sb,
    float* __restrict__ C, int M, int N, int K, int tiles_n) {
  constexpr int kABytes = 64 * 128;
  __shared__ __attribute__((aligned(16))) char smem[2 * 2 * kABytes];

  uint32_t bid = xcd_swizzle(blockIdx.x, gridDim.x);
  int m0 = (int)(bid / tiles_n) * 64;
  int n0 = (int)(bid % tiles_n) * 64;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int ktiles = K >> 7;       // 128 elems per tile
  const int kblocks = K >> 5;      // 32-elem MX blocks per row

  f32x4 acc[4];
#pragma unroll
  for (int f = 0; f < 4; ++f) acc[f] = {0.f, 0.f, 0.f, 0.f};

  auto stage = [&](int t, int slot) {
    char* base = smem + slot * 2 * kABytes;
    // A rows [m0, m0+64), B rows [n0, n0+64): 64 rows x 8 chunks each.
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = c * 256 + tid;  // 0..511
      uint32_t row = (uint32_t)idx >> 3;
      uint32_t cb = ((uint32_t)idx & 7) * 16;
      int ar = m0 + (int)row;
      if (ar >= M) ar = M - 1;
      glds16((const char*)(A + (int64_t)ar * K + t * 128 + cb),
             (uint32_t)(uintptr_t)base + row * 128 + (cb ^ ((row & 7) << 4)));
      int br = n0 + (int)row;
      if (br >= N) br = N - 1;
      glds16((const char*)(B + (int64_t)br * K + t * 128 + cb),
             (uint32_t)(uintptr_t)(base + kABytes) + row * 128 +
                 (cb ^ ((row & 7) << 4)));
    }
  };

  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  int cur = 0;
  int arow = m0 + wave * 16 + (lane & 15);
  if (arow >= M) arow = M - 1;
  for (int t = 0; t < ktiles; ++t) {
    if (t + 1 < ktiles) stage(t + 1, cur ^ 1);
    const char* As = smem + cur * 2 * kABytes;
    const char* Bs = As + kABytes;
    int kb = t * 4 + (lane >> 4);  // this lane's MX block index
    i32x8v af = read_frag32(As, wave * 16 + (lane & 15), lane);
    int sa = Sa[(int64_t)arow * kblocks + kb];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int brow = n0 + f * 16 + (lane & 15);
      if (brow >= N) brow = N - 1;
      i32x8v bf = read_frag32(Bs, f * 16 + (lane & 15), lane);
      int sb = Sb[(int64_t)brow * kblocks + kb];
      acc[f] = __builtin_amdgcn_mfma_scale_f32_16x16x128_f8f6f4(
          af, bf, acc[f], 0 /*cbsz: A fp8*/, 0 /*blgp: B fp8*/, 0, sa, 0, sb);
    }
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

#pragma unroll
  for (int f = 0; f < 4; ++f) {
    int col = n0 + f * 16 + (lane & 15);
    if (col >= N) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = m0 + wave * 16 + ((lane >> 4) << 2) + r;
      if (row >= M) continue;
      C[(int64_t)row * N + col] = acc[f][r];
    }
  }
}

void launch_gemm_mxfp8(const void* A, const void* B, const void* Sa,
                       const void* Sb, void* C, int M, int N, int K,
                       hipStream_t stream) {
  if (K % 128 != 0)
    throw std::runtime_error("gemm_mxfp8: K must be a multiple of 128");
  int tiles_n = (int)cdiv(N, 64);
  dim3 grid((unsigned)(cdiv(M, 64) * tiles_n));
  hipLaunchKernelGGL(gemm_mxfp8_kernel, grid, dim3(256), 0, stream,
                     (const uint8_t*)A, (const uint8_t*)B,
                     (const uint8_t*)Sa, (const uint8_t*)Sb, (float*)C, M, N,
                     K, tiles_n);
}

}  // namespace trtlab
