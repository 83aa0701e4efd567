// trtlab_amd — fused multi-head self-attention (gfx950). SURVEY.md §2.8
// item 3: batched GEMM + softmax fused in one kernel.
//
// Input: qkv [B*S, 3*H*D] (the fused QKV projection output, column blocks
// q|k|v, each H*D with head-major (h,d) minor d). Output: [B*S, H*D].
//
// One workgroup per (b, h, 64-query-row block); 4 waves own 16 query rows
// each. Keys/values stream through LDS in 128-key tiles with an ONLINE
// softmax (flash-attention style running max/sum with accumulator
// rescaling), so S is any multiple of 128 (BERT 128 ... GPT-2 1024) at
// fixed LDS footprint:
//   per key tile: QK^T via mfma_f32_16x16x32 (K tile is "bt" layout),
//   wave-local running softmax (rows on 16-lane groups; shfl_xor),
//   P staged to per-wave LDS, PV accumulated against V^T (staged
//   transposed at load), output rescaled by exp(m_old - m_new).
// fp32 accumulation; scores scaled by 1/sqrt(D).
//
// Generality: S is ARBITRARY (tail query rows clamp
// their loads and skip their stores; tail key tiles clamp loads and mask
// scores >= S), and D in {64, 128} — the D=128 variant below retiles to
// 64-key tiles so Q[64][128] + K[64][128] + Vt[128][64] + P fit in 56 KiB
// of LDS.
//
// Masks: seqlens (optional, [B] device ints) masks right-padded keys
// >= seqlens[b] (variable-length batches; key tiles past the valid length
// are skipped entirely). causal != 0 masks keys > query (decoder-style),
// and skips key tiles entirely above the block's query range.
#include "gemm_common.h"

namespace trtlab {

// OT: output element type (fp8 e4m3 with out_scale = 1/s_q fuses the
// producer-side quantization for the following projection GEMM).
template <typename T, typename OT = T>
__global__ __launch_bounds__(256) void attention_kernel(
    const T* __restrict__ qkv, OT* __restrict__ out, int B, int S, int H,
    int D, float scale, float out_scale, const int* __restrict__ seqlens,
    int causal) {
  // LDS: Q [64][64] | K [128][64] | Vt 2x[64][64] | P 4 waves x 2x[16][64]
  __shared__ __attribute__((aligned(16))) char smem[8192 + 16384 * 2 + 16384];
  char* Qs = smem;                  // 8 KiB
  char* Ks = smem + 8192;          // 16 KiB
  char* Vt = smem + 8192 + 16384;  // two [64][64] tiles: kt*8192
  char* Ps = Vt + 16384;           // per wave: wave*4096, two [16][64] tiles

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int qblocks = (S + 63) >> 6;  // 64-query-row blocks per (b, h)
  const int blk = blockIdx.x;
  const int bh = blk / qblocks;
  const int q0 = (blk % qblocks) * 64;  // this block's query-row base
  const int b = bh / H;
  const int h = bh % H;
  const int hid = H * D;
  const int row_stride = 3 * hid;

  const T* base = qkv + (int64_t)b * S * row_stride;
  const int qoff = h * D;
  const int koff = hid + h * D;
  const int voff = 2 * hid + h * D;

  auto swz = [](uint32_t row, uint32_t colbyte) {
    return row * 128 + (colbyte ^ ((row & 7) << 4));
  };

  // ---- stage Q once (64 rows x 8 chunks = 512 chunks, 2/thread) ----
  // tail blocks clamp out-of-range query rows (their scores are computed
  // but never stored)
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    int idx = c * 256 + tid;       // 0..511
    int row = idx >> 3;            // 0..63
    int cb = (idx & 7) * 16;
    int qr = q0 + row < S ? q0 + row : S - 1;
    const T* src = base + (int64_t)qr * row_stride + qoff + cb / 2;
    *(short8v*)(Qs + swz(row, cb)) = *(const short8v*)src;
  }

  int limit = S;
  if (seqlens) {
    limit = seqlens[b];
    limit = limit < 1 ? 1 : (limit > S ? S : limit);
  }
  int ntiles = (limit + 127) >> 7;          // key tiles with any valid key
  if (causal) {
    int tmax = (q0 + 64 + 127) >> 7;        // keys above the block's queries
    ntiles = ntiles < tmax ? ntiles : tmax; // contribute nothing
  }

  using MF = Mfma16x16x32<T>;
  const int qrow = wave * 16;      // within the block's 64-row slice
  float m_run[4], l_run[4];
  f32x4 oacc[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -3.0e38f;
    l_run[r] = 0.f;
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) oacc[j] = {0.f, 0.f, 0.f, 0.f};

  // K/V staging is register-double-buffered: tile t+1's global loads are
  // issued while tile t computes (their ~1-2 us latency was serialized
  // with compute before), and only the cheap ds_writes sit between the
  // barriers. LDS footprint unchanged (occupancy stays 2/SIMD).
  short8v kv_k[4], kv_v[4];
  auto load_kv = [&](int t) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int idx = c * 256 + tid;
      int row = idx >> 3;            // 0..127
      int cb = (idx & 7) * 16;
      int kr = t * 128 + row < S ? t * 128 + row : S - 1;  // tail clamp
      kv_k[c] = *(const short8v*)(
          base + (int64_t)kr * row_stride + koff + cb / 2);
      int ci = c * 256 + tid;        // 0..1023
      int key = ci >> 3;
      int dd = (ci & 7) * 8;
      int vr = t * 128 + key < S ? t * 128 + key : S - 1;
      kv_v[c] = *(const short8v*)(
          base + (int64_t)vr * row_stride + voff + dd);
    }
  };
  if (ntiles > 0) load_kv(0);
  for (int t = 0; t < ntiles; ++t) {
    // previous tile's PV reads must finish before K/Vt are overwritten
    __syncthreads();
    // ---- write the prefetched K tile (128 rows) + Vt (transposed) ----
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int idx = c * 256 + tid;
      int row = idx >> 3;
      int cb = (idx & 7) * 16;
      *(short8v*)(Ks + swz(row, cb)) = kv_k[c];
      int ci = c * 256 + tid;
      int key = ci >> 3;
      int dd = (ci & 7) * 8;
      char* tile = Vt + (key >> 6) * 8192;
      int kcol = key & 63;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        *(T*)(tile + swz(dd + j, kcol * 2)) = ((const T*)&kv_v[c])[j];
      }
    }
    if (t + 1 < ntiles) load_kv(t + 1);  // overlaps this tile's compute
    __syncthreads();

    // ---- QK^T for this wave's 16 query rows ----
    f32x4 sacc[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) sacc[j] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      uint32_t kbyte = ks * 64 + ((lane >> 4) << 4);
      typename MF::frag qf =
          *(const typename MF::frag*)(Qs + swz(qrow + (lane & 15), kbyte));
      typename MF::frag kf[8];
#pragma unroll
      for (int f = 0; f < 8; ++f)
        kf[f] =
            *(const typename MF::frag*)(Ks + swz(f * 16 + (lane & 15), kbyte));
#pragma unroll
      for (int j = 0; j < 8; ++j) sacc[j] = MF::run(qf, kf[j], sacc[j]);
    }

    // ---- masking ----
    // score fragment layout: key = t*128 + j*16 + (lane & 15) (MFMA C col),
    // query row = q0 + qrow + (lane>>4)*4 + r.
    if (limit - t * 128 < 128) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        if (t * 128 + j * 16 + (lane & 15) >= limit) {
#pragma unroll
          for (int r = 0; r < 4; ++r) sacc[j][r] = -3.0e38f;
        }
      }
    }
    if (causal && t * 128 + 127 > q0 + qrow) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int key = t * 128 + j * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int q = q0 + qrow + ((lane >> 4) << 2) + r;
          if (key > q) sacc[j][r] = -3.0e38f;
        }
      }
    }

    // ---- online softmax update (rows on 16-lane groups; shfl_xor) ----
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m = m_run[r];
#pragma unroll
      for (int j = 0; j < 8; ++j) m = fmaxf(m, sacc[j][r] * scale);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        m = fmaxf(m, __shfl_xor(m, off, 64));
      float c = __expf(m_run[r] - m);  // 0 on the first tile (m_run = -inf)
      float s = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float e = __expf(sacc[j][r] * scale - m);
        sacc[j][r] = e;
        s += e;
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) s += __shfl_xor(s, off, 64);
      l_run[r] = l_run[r] * c + s;
      m_run[r] = m;
#pragma unroll
      for (int j = 0; j < 4; ++j) oacc[j][r] *= c;
    }

    // ---- P (unnormalized exp weights) -> per-wave LDS, two [16][64] ----
    char* P = Ps + wave * 4096;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int colg = j * 16 + (lane & 15);
      char* tile = P + (colg >> 6) * 2048;
      int col = colg & 63;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = ((lane >> 4) << 2) + r;  // 0..15
        *(T*)(tile + ((uint32_t)row * 128 + ((col * 2) ^ ((row & 7) << 4)))) =
            (T)sacc[j][r];
      }
    }
    // Make the P ds_writes visible before the PV ds_reads (same wave, but
    // the explicit barrier proved necessary — see git history).
    __syncthreads();

    // ---- PV: oacc[16 rows][64 d] += P[16][128] @ Vt^T ----
#pragma unroll
    for (int kt = 0; kt < 2; ++kt) {
      const char* Pt = P + kt * 2048;
      const char* Vk = Vt + kt * 8192;
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        uint32_t kbyte = ks * 64 + ((lane >> 4) << 4);
        typename MF::frag pf =
            *(const typename MF::frag*)(Pt + swz(lane & 15, kbyte));
        typename MF::frag vf[4];
#pragma unroll
        for (int f = 0; f < 4; ++f)
          vf[f] = *(const typename MF::frag*)(Vk +
                   swz(f * 16 + (lane & 15), kbyte));
#pragma unroll
        for (int j = 0; j < 4; ++j) oacc[j] = MF::run(pf, vf[j], oacc[j]);
      }
    }
  }

  // ---- store out[b*S + q0 + row][h*D + d] = oacc / l ----
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int d = j * 16 + (lane & 15);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = q0 + qrow + ((lane >> 4) << 2) + r;
      if (row >= S) continue;  // tail block: clamped rows are not stored
      float inv = l_run[r] > 0.f ? 1.0f / l_run[r] : 0.f;
      out[((int64_t)b * S + row) * hid + h * D + d] =
          store_cast<OT>(oacc[j][r] * inv * out_scale);
    }
  }
}

// ---- D = 128 variant (head_dim-128 decoders / LLaMA-family shapes) ----
// Retiled for the bigger head: 64-key tiles keep the whole working set in
// 56 KiB LDS (Q[64][128] 16K + K[64][128] 16K + Vt[128][64] 16K + P 8K).
// Same online softmax, masking, clamped-tail generality as the D=64 path.
template <typename T, typename OT = T>
__global__ __launch_bounds__(256) void attention_kernel_d128(
    const T* __restrict__ qkv, OT* __restrict__ out, int B, int S, int H,
    float scale, float out_scale, const int* __restrict__ seqlens,
    int causal) {
  constexpr int D = 128;
  __shared__ __attribute__((aligned(16))) char smem[16384 * 3 + 8192];
  char* Qs = smem;                   // [64 q][128 d], 256-B rows
  char* Ks = smem + 16384;           // [64 k][128 d], 256-B rows
  char* Vt = smem + 32768;           // [128 d][64 k], 128-B rows
  char* Ps = smem + 49152;           // per wave [16 q][64 k], 128-B rows

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int qblocks = (S + 63) >> 6;
  const int blk = blockIdx.x;
  const int bh = blk / qblocks;
  const int q0 = (blk % qblocks) * 64;
  const int b = bh / H;
  const int h = bh % H;
  const int hid = H * D;
  const int row_stride = 3 * hid;

  const T* base = qkv + (int64_t)b * S * row_stride;
  const int qoff = h * D;
  const int koff = hid + h * D;
  const int voff = 2 * hid + h * D;

  // 256-B-row swizzle: 16 distinct 16-B offsets per 16 rows (bank period
  // = 256 B on 64x4-B LDS banks) -> conflict-free frag reads and writes
  auto swz128 = [](uint32_t row, uint32_t colbyte) {
    return row * 256 + (colbyte ^ ((row & 15) << 4));
  };
  auto swz = [](uint32_t row, uint32_t colbyte) {  // 128-B rows (Vt, P)
    return row * 128 + (colbyte ^ ((row & 7) << 4));
  };

  // ---- stage Q once: 64 rows x 16 chunks = 1024 chunks, 4/thread ----
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    int idx = c * 256 + tid;
    int row = idx >> 4;              // 0..63
    int cb = (idx & 15) * 16;
    int qr = q0 + row < S ? q0 + row : S - 1;
    *(short8v*)(Qs + swz128(row, cb)) =
        *(const short8v*)(base + (int64_t)qr * row_stride + qoff + cb / 2);
  }

  int limit = S;
  if (seqlens) {
    limit = seqlens[b];
    limit = limit < 1 ? 1 : (limit > S ? S : limit);
  }
  int ntiles = (limit + 63) >> 6;           // 64-key tiles
  if (causal) {
    int tmax = (q0 + 64 + 63) >> 6;
    ntiles = ntiles < tmax ? ntiles : tmax;
  }

  using MF = Mfma16x16x32<T>;
  const int qrow = wave * 16;
  float m_run[4], l_run[4];
  f32x4 oacc[8];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -3.0e38f;
    l_run[r] = 0.f;
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) oacc[j] = {0.f, 0.f, 0.f, 0.f};

  short8v kv_k[4], kv_v[4];
  auto load_kv = [&](int t) {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int idx = c * 256 + tid;
      int row = idx >> 4;            // 0..63 (key)
      int cb = (idx & 15) * 16;
      int kr = t * 64 + row < S ? t * 64 + row : S - 1;
      kv_k[c] = *(const short8v*)(
          base + (int64_t)kr * row_stride + koff + cb / 2);
      int key = idx >> 4;
      int dd = (idx & 15) * 8;
      int vr = t * 64 + key < S ? t * 64 + key : S - 1;
      kv_v[c] = *(const short8v*)(
          base + (int64_t)vr * row_stride + voff + dd);
    }
  };
  if (ntiles > 0) load_kv(0);
  for (int t = 0; t < ntiles; ++t) {
    __syncthreads();
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      int idx = c * 256 + tid;
      int row = idx >> 4;
      int cb = (idx & 15) * 16;
      *(short8v*)(Ks + swz128(row, cb)) = kv_k[c];
      int key = idx >> 4;
      int dd = (idx & 15) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        *(T*)(Vt + swz(dd + j, key * 2)) = ((const T*)&kv_v[c])[j];
    }
    if (t + 1 < ntiles) load_kv(t + 1);
    __syncthreads();

    // ---- QK^T: 16 q rows x 64 keys, k-dim 128 (4 MFMA k-steps) ----
    f32x4 sacc[4];
#pragma unroll
    for (int j = 0; j < 4; ++j) sacc[j] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      uint32_t kbyte = ks * 64 + ((lane >> 4) << 4);
      typename MF::frag qf = *(const typename MF::frag*)(
          Qs + swz128(qrow + (lane & 15), kbyte));
      typename MF::frag kf[4];
#pragma unroll
      for (int f = 0; f < 4; ++f)
        kf[f] = *(const typename MF::frag*)(
            Ks + swz128(f * 16 + (lane & 15), kbyte));
#pragma unroll
      for (int j = 0; j < 4; ++j) sacc[j] = MF::run(qf, kf[j], sacc[j]);
    }

    // ---- masking (key = t*64 + j*16 + (lane & 15)) ----
    if (limit - t * 64 < 64) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        if (t * 64 + j * 16 + (lane & 15) >= limit) {
#pragma unroll
          for (int r = 0; r < 4; ++r) sacc[j][r] = -3.0e38f;
        }
      }
    }
    if (causal && t * 64 + 63 > q0 + qrow) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int key = t * 64 + j * 16 + (lane & 15);
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int q = q0 + qrow + ((lane >> 4) << 2) + r;
          if (key > q) sacc[j][r] = -3.0e38f;
        }
      }
    }

    // ---- online softmax ----
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m = m_run[r];
#pragma unroll
      for (int j = 0; j < 4; ++j) m = fmaxf(m, sacc[j][r] * scale);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        m = fmaxf(m, __shfl_xor(m, off, 64));
      float c = __expf(m_run[r] - m);
      float s = 0.f;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float e = __expf(sacc[j][r] * scale - m);
        sacc[j][r] = e;
        s += e;
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) s += __shfl_xor(s, off, 64);
      l_run[r] = l_run[r] * c + s;
      m_run[r] = m;
#pragma unroll
      for (int j = 0; j < 8; ++j) oacc[j][r] *= c;
    }

    // ---- P -> per-wave LDS [16 q][64 k] ----
    char* P = Ps + wave * 2048;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = j * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = ((lane >> 4) << 2) + r;
        *(T*)(P + ((uint32_t)row * 128 + ((col * 2) ^ ((row & 7) << 4)))) =
            (T)sacc[j][r];
      }
    }
    __syncthreads();

    // ---- PV: oacc[16 q][128 d] += P[16][64] @ Vt^T ----
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      uint32_t kbyte = ks * 64 + ((lane >> 4) << 4);
      typename MF::frag pf =
          *(const typename MF::frag*)(P + swz(lane & 15, kbyte));
      typename MF::frag vf[8];
#pragma unroll
      for (int f = 0; f < 8; ++f)
        vf[f] = *(const typename MF::frag*)(
            Vt + swz(f * 16 + (lane & 15), kbyte));
#pragma unroll
      for (int j = 0; j < 8; ++j) oacc[j] = MF::run(pf, vf[j], oacc[j]);
    }
  }

  // ---- store ----
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int d = j * 16 + (lane & 15);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = q0 + qrow + ((lane >> 4) << 2) + r;
      if (row >= S) continue;
      float inv = l_run[r] > 0.f ? 1.0f / l_run[r] : 0.f;
      out[((int64_t)b * S + row) * hid + h * D + d] =
          store_cast<OT>(oacc[j][r] * inv * out_scale);
    }
  }
}

// Debug probe: each block writes its view of the launch parameters.
__global__ void attention_probe_kernel(int* dbg, int B, int S, int H, int D) {
  if (threadIdx.x == 0) {
    int bh = blockIdx.x;
    int* p = dbg + bh * 8;
    p[0] = B; p[1] = S; p[2] = H; p[3] = D;
    p[4] = (int)gridDim.x; p[5] = bh; p[6] = bh / H; p[7] = bh % H;
  }
}

void launch_attention_probe(int* dbg, int B, int S, int H, int D,
                            hipStream_t stream) {
  hipLaunchKernelGGL(attention_probe_kernel, dim3(B * H), dim3(256), 0,
                     stream, dbg, B, S, H, D);
}

void launch_attention(int dtype, const void* qkv, void* out, int B, int S,
                      int H, int D, float scale, hipStream_t stream,
                      int out_dtype, float out_scale, const void* seqlens,
                      int causal) {
  if (D != 64 && D != 128)
    throw std::runtime_error("attention: head_dim must be 64 or 128");
  if (S < 1) throw std::runtime_error("attention: S must be >= 1");
  dim3 grid(B * H * ((S + 63) / 64));  // 64-query-row blocks (+ tail)
  dim3 block(256);
  const int* lens = (const int*)seqlens;
  if (D == 128) {
    if (dtype == 0) {
      if (out_dtype == 3)
        hipLaunchKernelGGL((attention_kernel_d128<_Float16, __hip_fp8_e4m3>),
                           grid, block, 0, stream, (const _Float16*)qkv,
                           (__hip_fp8_e4m3*)out, B, S, H, scale, out_scale,
                           lens, causal);
      else
        hipLaunchKernelGGL((attention_kernel_d128<_Float16, _Float16>), grid,
                           block, 0, stream, (const _Float16*)qkv,
                           (_Float16*)out, B, S, H, scale, out_scale, lens,
                           causal);
    } else {
      hipLaunchKernelGGL((attention_kernel_d128<__bf16, __bf16>), grid, block,
                         0, stream, (const __bf16*)qkv, (__bf16*)out, B, S, H,
                         scale, out_scale, lens, causal);
    }
    return;
  }
  if (dtype == 0) {
    if (out_dtype == 3)  // fused fp8 output for the projection GEMM
      hipLaunchKernelGGL((attention_kernel<_Float16, __hip_fp8_e4m3>), grid,
                         block, 0, stream, (const _Float16*)qkv,
                         (__hip_fp8_e4m3*)out, B, S, H, D, scale, out_scale,
                         lens, causal);
    else
      hipLaunchKernelGGL((attention_kernel<_Float16, _Float16>), grid, block,
                         0, stream, (const _Float16*)qkv, (_Float16*)out, B,
                         S, H, D, scale, out_scale, lens, causal);
  } else {
    hipLaunchKernelGGL((attention_kernel<__bf16, __bf16>), grid, block, 0,
                       stream, (const __bf16*)qkv, (__bf16*)out, B, S, H, D,
                       scale, out_scale, lens, causal);
  }
}

// ---- seqlens from right-padded token ids ----
// lens[b] = count of ids[b*S + i] != pad_id (clamped to >= 1). One wave
// per sequence; the planner schedules this once per forward when any
// attention op runs in variable-length mode.
__global__ __launch_bounds__(64) void seqlens_kernel(
    const int* __restrict__ ids, int* __restrict__ lens, int S, int pad_id) {
  int b = blockIdx.x;
  int lane = threadIdx.x;
  int cnt = 0;
  for (int i = lane; i < S; i += 64) cnt += (ids[b * S + i] != pad_id);
#pragma unroll
  for (int off = 32; off; off >>= 1) cnt += __shfl_down(cnt, off, 64);
  if (lane == 0) lens[b] = cnt > 0 ? cnt : 1;
}

void launch_seqlens(const void* ids, void* lens, int B, int S, int pad_id,
                    hipStream_t stream) {
  hipLaunchKernelGGL(seqlens_kernel, dim3(B), dim3(64), 0, stream,
                     (const int*)ids, (int*)lens, S, pad_id);
}

}  // namespace trtlab
