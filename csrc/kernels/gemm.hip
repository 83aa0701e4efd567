// trtlab_amd — MFMA GEMM for gfx950 (CDNA4), fp16/bf16 in, fp32 accumulate.
//
// C[M][N] = epilogue( A[M][K] @ B[N][K]^T )
//
// B is stored transposed ([N][K], "bt") — weights are prepacked at engine
// build time into this layout so both LDS tiles stage identically and every
// MFMA fragment read is a contiguous ds_read_b128.
//
// Structure (cdna_hip_programming.md §5, "minimum 2-phase" recipe):
//   - BMxBN output tile (picked per shape: 128x128 .. 64x64 so small
//     deep-layer shapes still fill 256 CUs), BK=64, 4 waves (2x2).
//   - global->LDS staging via __builtin_amdgcn_global_load_lds (16 B/lane),
//     double-buffered; one vmcnt(0) + barrier per K-tile.
//   - LDS XOR swizzle ((row&7)<<4) applied on the SOURCE address and
//     re-applied on every ds_read_b128 (both-sides-or-neither, rule 21).
//   - XCD-aware bijective blockIdx swizzle (T1); a split tile's K-slices
//     stay adjacent (same XCD) per the split-K guidance.
//   - split-K for K-heavy grid-starved shapes: fp32 slabs + a deterministic
//     reduce kernel (splitk.hip) that applies the epilogue.
//
// Replaces the TensorRT-internal GEMM path of the reference
// (trtlab/tensorrt/src/workspace.cc:47 enqueueV2 — opaque engine kernels).
#include "gemm_common.h"

namespace trtlab {

void launch_splitk_reduce(int dtype, const float* scratch, void* C,
                          const float* scale, const float* bias,
                          const void* residual, float res_scale, int M, int N,
                          int64_t ldc, int tiles_m, int tiles_n, int splitk,
                          int bm, int bn, int epi, hipStream_t stream);

template <typename T, typename OT, Epi E, int BM, int BN, bool SPLIT,
          int NBUF>
__global__ __launch_bounds__(256) void gemm_bt_kernel(
    const T* __restrict__ A, const T* __restrict__ B, OT* __restrict__ C,
    const float* __restrict__ scale, const float* __restrict__ bias,
    const OT* __restrict__ residual, float res_scale, float out_scale,
    int M, int N, int K, int64_t lda, int64_t ldb, int64_t ldc, int tiles_n,
    float* __restrict__ scratch, int splitk, int ktper) {
  constexpr int kABytes = BM * 128;
  constexpr int kBuf = (BM + BN) * 128;

  uint32_t bid = xcd_swizzle(blockIdx.x, gridDim.x);
  uint32_t tile = SPLIT ? bid / splitk : bid;
  int m0 = (int)(tile / tiles_n) * BM;
  int n0 = (int)(tile % tiles_n) * BN;

  constexpr int KT = kTileElems<T>;  // 64 (fp16/bf16) or 128 (int8)
  const int ktiles = K / KT;
  int kt0 = 0, kt1 = ktiles;
  if constexpr (SPLIT) {
    int slice = bid % splitk;
    kt0 = slice * ktper;
    kt1 = min(ktiles, kt0 + ktper);
  }

  __shared__ __attribute__((aligned(16))) char smem[NBUF * kBuf];
  uint32_t lds0 = (uint32_t)(uintptr_t)&smem[0];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;

  typename Mfma16x16x32<T>::accv acc[BM / 32][BN / 32];
#pragma unroll
  for (int i = 0; i < BM / 32; ++i)
#pragma unroll
    for (int j = 0; j < BN / 32; ++j) acc[i][j] = {0, 0, 0, 0};

  auto stage = [&](int t, int slot) {
    uint32_t base = lds0 + slot * kBuf;
    stage_tile<T, BM>(A + (int64_t)m0 * lda + (int64_t)t * KT, lda, m0, M,
                      base, tid);
    stage_tile<T, BN>(B + (int64_t)n0 * ldb + (int64_t)t * KT, ldb, n0, N,
                      base + kABytes, tid);
  };

  if constexpr (NBUF == 4) {
    // 3-deep staging pipeline: counted vmcnt + raw barrier (loads stay in
    // flight across barriers; see gemm_common.h wait_tiles_inflight).
    constexpr int G = BM / 32 + BN / 32;
    for (int i = 0; i < 3 && kt0 + i < kt1; ++i) stage(kt0 + i, i);
    for (int t = kt0; t < kt1; ++t) {
      int ahead = kt1 - 1 - t;
      if (ahead > 2) ahead = 2;
      wait_tiles_inflight<G>(ahead);
      __builtin_amdgcn_s_barrier();
      if (t + 3 < kt1) stage(t + 3, (t + 3 - kt0) & 3);
      const char* As = &smem[((t - kt0) & 3) * kBuf];
      mfma_tile<T, BM, BN>(As, As + kABytes, lane, wr, wc, acc);
    }
  } else if constexpr (NBUF == 3) {
    // 2-deep counted pipeline (the 256-wide tile: 4 slots would overflow
    // LDS; occupancy is 1, so in-block overlap does the latency hiding)
    constexpr int G = BM / 32 + BN / 32;
    stage(kt0, 0);
    if (kt0 + 1 < kt1) stage(kt0 + 1, 1);
    for (int t = kt0; t < kt1; ++t) {
      int ahead = kt1 - 1 - t;
      if (ahead > 1) ahead = 1;
      wait_tiles_inflight<G>(ahead);
      __builtin_amdgcn_s_barrier();
      if (t + 2 < kt1) stage(t + 2, (t + 2 - kt0) % 3);
      const char* As = &smem[((t - kt0) % 3) * kBuf];
      mfma_tile<T, BM, BN>(As, As + kABytes, lane, wr, wc, acc);
    }
  } else {
    stage(kt0, 0);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    int cur = 0;
    for (int t = kt0; t < kt1; ++t) {
      if (t + 1 < kt1) stage(t + 1, cur ^ 1);
      const char* As = &smem[cur * kBuf];
      mfma_tile<T, BM, BN>(As, As + kABytes, lane, wr, wc, acc);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
      cur ^= 1;
    }
  }

  if constexpr (SPLIT) {
    store_splitk<T, BM, BN>(acc, scratch + (int64_t)bid * BM * BN, lane, wr,
                            wc);
  } else {
    store_epilogue<T, E, BM, BN, OT>(acc, C, ldc, m0, n0, M, N, scale, bias,
                                     residual, res_scale, lane, wr, wc,
                                     out_scale);
  }
}

size_t gemm_scratch_bytes(int M, int N, int K) {
  TileCfg cfg = pick_tile(M, N);
  long tiles = cdiv(M, cfg.bm) * cdiv(N, cfg.bn);
  int splitk = pick_splitk_gemm(tiles, K >> 6);  // fp16 tiles (conservative)
  if (splitk == 1) return 0;
  return (size_t)tiles * splitk * cfg.bm * cfg.bn * 4;
}

template <typename T, typename OT = T>
static void launch_gemm_bt_t(const void* A, const void* B, void* C,
                             const float* scale, const float* bias,
                             const void* residual, float res_scale,
                             float out_scale, int M, int N, int K,
                             int64_t lda, int64_t ldb, int64_t ldc, int epi,
                             hipStream_t stream, int tile, float* scratch) {
  TileCfg cfg = tile ? tile_from_code(tile) : pick_tile(M, N);
  int tiles_m = (int)cdiv(M, cfg.bm);
  int tiles_n = (int)cdiv(N, cfg.bn);
  long tiles = (long)tiles_m * tiles_n;
  int ktiles = K / kTileElems<T>;
  if (cfg.bm == 256) {
    // 256x128 tactic (code 5): 3-slot counted pipeline (144 KiB LDS),
    // no split-K; picked by autotune on staging-bound large-M shapes.
    dim3 grid5((unsigned)tiles);
    epi_dispatch(epi, [&](auto e) {
      constexpr Epi EE = decltype(e)::value;
      hipLaunchKernelGGL((gemm_bt_kernel<T, OT, EE, 256, 128, false, 3>),
                         grid5, dim3(256), 0, stream, (const T*)A,
                         (const T*)B, (OT*)C, scale, bias,
                         (const OT*)residual, res_scale, out_scale, M, N, K,
                         lda, ldb, ldc, tiles_n, (float*)nullptr, 1, ktiles);
    });
    return;
  }
  int splitk = (!tile && scratch) ? pick_splitk_gemm(tiles, ktiles) : 1;
  dim3 block(256);
  int out_dtype = std::is_same<OT, _Float16>::value
                      ? 0
                      : (std::is_same<OT, __bf16>::value
                             ? 1
                             : (std::is_same<OT, int8_t>::value ? 2 : 3));
  if (splitk > 1) {
    int ktper = (int)cdiv(ktiles, splitk);
    dim3 grid((unsigned)(tiles * splitk));
    bool deep = want_deep_pipe(tiles * splitk, ktper);
    tile_dispatch(cfg, [&](auto bm, auto bn) {
      constexpr int BM = decltype(bm)::value;
      constexpr int BN = decltype(bn)::value;
      if (deep)
        hipLaunchKernelGGL((gemm_bt_kernel<T, OT, Epi::kNone, BM, BN, true, 4>),
                           grid, block, 0, stream, (const T*)A, (const T*)B,
                           (OT*)C, scale, bias, (const OT*)residual,
                           res_scale, out_scale, M, N, K, lda, ldb, ldc,
                           tiles_n, scratch, splitk, ktper);
      else
        hipLaunchKernelGGL((gemm_bt_kernel<T, OT, Epi::kNone, BM, BN, true, 2>),
                           grid, block, 0, stream, (const T*)A, (const T*)B,
                           (OT*)C, scale, bias, (const OT*)residual,
                           res_scale, out_scale, M, N, K, lda, ldb, ldc,
                           tiles_n, scratch, splitk, ktper);
    });
    launch_splitk_reduce(out_dtype, scratch, C, scale, bias, residual,
                         res_scale, M, N, ldc, tiles_m, tiles_n, splitk,
                         cfg.bm, cfg.bn, epi, stream);
    return;
  }
  dim3 grid((unsigned)tiles);
  bool deep = want_deep_pipe(tiles, ktiles);
  epi_dispatch(epi, [&](auto e) {
    constexpr Epi EE = decltype(e)::value;
    tile_dispatch(cfg, [&](auto bm, auto bn) {
      constexpr int BM = decltype(bm)::value;
      constexpr int BN = decltype(bn)::value;
      if (deep)
        hipLaunchKernelGGL((gemm_bt_kernel<T, OT, EE, BM, BN, false, 4>),
                           grid, block, 0, stream, (const T*)A, (const T*)B,
                           (OT*)C, scale, bias, (const OT*)residual,
                           res_scale, out_scale, M, N, K, lda, ldb, ldc,
                           tiles_n, (float*)nullptr, 1, ktiles);
      else
        hipLaunchKernelGGL((gemm_bt_kernel<T, OT, EE, BM, BN, false, 2>),
                           grid, block, 0, stream, (const T*)A, (const T*)B,
                           (OT*)C, scale, bias, (const OT*)residual,
                           res_scale, out_scale, M, N, K, lda, ldb, ldc,
                           tiles_n, (float*)nullptr, 1, ktiles);
    });
  });
}

void launch_gemm_bt(int dtype,  // 0 fp16, 1 bf16, 2 int8, 3 fp8, 4 fp8->fp16
                    const void* A, const void* B, void* C, const float* scale,
                    const float* bias, const void* residual, float res_scale,
                    int M, int N, int K, int64_t lda, int64_t ldb, int64_t ldc,
                    int epi, hipStream_t stream, int tile, void* scratch,
                    float out_scale) {
  if (dtype == 2 || dtype == 3 || dtype == 4) {
    if (K % 128 != 0)
      throw std::runtime_error("gemm_bt int8/fp8: K must be a multiple of 128");
    // the split-K reduce path does not apply out_scale; disable split
    // when a post-activation quant scale is present
    void* sk = out_scale == 1.0f ? scratch : nullptr;
    if (dtype == 2)
      launch_gemm_bt_t<int8_t>(A, B, C, scale, bias, residual, res_scale,
                               out_scale, M, N, K, lda, ldb, ldc, epi,
                               stream, tile, (float*)sk);
    else if (dtype == 3)
      launch_gemm_bt_t<__hip_fp8_e4m3>(A, B, C, scale, bias, residual,
                                       res_scale, out_scale, M, N, K, lda,
                                       ldb, ldc, epi, stream, tile,
                                       (float*)sk);
    else  // 4: fp8 compute, fp16 output (transformer projections)
      launch_gemm_bt_t<__hip_fp8_e4m3, _Float16>(
          A, B, C, scale, bias, residual, res_scale, out_scale, M, N, K, lda,
          ldb, ldc, epi, stream, tile, (float*)sk);
    return;
  }
  if (K % 64 != 0) throw std::runtime_error("gemm_bt: K must be a multiple of 64");
  if (dtype == 0)
    launch_gemm_bt_t<_Float16>(A, B, C, scale, bias, residual, res_scale,
                               out_scale, M, N, K, lda, ldb, ldc, epi,
                               stream, tile, (float*)scratch);
  else
    launch_gemm_bt_t<__bf16>(A, B, C, scale, bias, residual, res_scale,
                             out_scale, M, N, K, lda, ldb, ldc, epi, stream,
                             tile, (float*)scratch);
}

}  // namespace trtlab
