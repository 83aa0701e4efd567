// trtlab_amd — incremental (KV-cache) decode kernels for GPT-family
// serving on gfx950. Beyond-reference capability: the CUDA reference
// served static TensorRT engines only.
//
// Design: the decode step is captured in a hipGraph once and replayed per
// token. All position dependence goes through a DEVICE-side PER-SEQUENCE
// counter array (`pos[B]`), so replays need no re-capture — and a slot
// can be reset to position 0 between replays (continuous batching in
// lockstep: a finished sequence's slot restarts a new request while the
// other slots keep decoding). advance_pos bumps every slot at the end of
// the captured step.
//
// Idle-slot masking: pos[b] < 0 marks slot b IDLE — every per-slot kernel
// early-exits for it (no cache writes, no attention scan, position stays
// parked), so empty slots in a continuous batch cost ~nothing while the
// captured graph keeps its fixed shape.
//
// Cache layout per layer: K and V as [B][H][Smax][64] fp16 (contiguous
// 128-B rows per key — one cacheline).
#include "gemm_common.h"

namespace trtlab {

// Scatter this step's K/V head rows from the fused qkv projection output
// (qkv [B, 3*H*64], one row per sequence) into the caches at `pos`.
__global__ __launch_bounds__(64) void kv_append_kernel(
    const _Float16* __restrict__ qkv, _Float16* __restrict__ kcache,
    _Float16* __restrict__ vcache, const int* __restrict__ pos, int B, int H,
    int smax, int D) {
  int b = blockIdx.x / H, h = blockIdx.x % H;
  int p = pos[b];
  if (p < 0) return;  // idle slot
  int hid = H * D;
  for (int d = threadIdx.x; d < D; d += 64) {
    int64_t src = (int64_t)b * 3 * hid + h * D + d;
    int64_t dst = (((int64_t)b * H + h) * smax + p) * D + d;
    kcache[dst] = qkv[src + hid];
    vcache[dst] = qkv[src + 2 * hid];
  }
}

void launch_kv_append(const void* qkv, void* kcache, void* vcache,
                      const void* pos, int B, int H, int smax,
                      hipStream_t stream, int D) {
  hipLaunchKernelGGL(kv_append_kernel, dim3(B * H), dim3(64), 0, stream,
                     (const _Float16*)qkv, (_Float16*)kcache,
                     (_Float16*)vcache, (const int*)pos, B, H, smax, D);
}

// Batch-scatter a full prompt's K/V head rows (qkv [B*P, 3*H*64], rows
// ordered b-major then position) into the caches at positions [0, P).
// Grid B*H*P; fused prefill runs the prompt through the full-sequence
// kernels once, then decode continues from position P.
__global__ __launch_bounds__(64) void kv_append_range_kernel(
    const _Float16* __restrict__ qkv, _Float16* __restrict__ kcache,
    _Float16* __restrict__ vcache, int B, int H, int P, int smax, int D) {
  int p = blockIdx.x % P;
  int bh = blockIdx.x / P;
  int b = bh / H, h = bh % H;
  int hid = H * D;
  for (int d = threadIdx.x; d < D; d += 64) {
    int64_t src = ((int64_t)b * P + p) * 3 * hid + h * D + d;
    int64_t dst = (((int64_t)b * H + h) * smax + p) * D + d;
    kcache[dst] = qkv[src + hid];
    vcache[dst] = qkv[src + 2 * hid];
  }
}

void launch_kv_append_range(const void* qkv, void* kcache, void* vcache,
                            int B, int H, int P, int smax,
                            hipStream_t stream, int D) {
  hipLaunchKernelGGL(kv_append_range_kernel, dim3(B * H * P), dim3(64), 0,
                     stream, (const _Float16*)qkv, (_Float16*)kcache,
                     (_Float16*)vcache, B, H, P, smax, D);
}

// Single-query attention against the cache: out[b, h*64+d] =
// softmax(q . K[0..pos]) @ V[0..pos]. One wave per (b, h); each lane owns
// keys lane, lane+64, ... for the score pass (its K rows are whole
// 128-byte cachelines), then output element d = lane for the PV pass with
// the probabilities broadcast through LDS. Smax <= 4096.
template <int D>  // head_dim 64 or 128
__global__ __launch_bounds__(64) void decode_attention_kernel(
    const _Float16* __restrict__ qkv, const _Float16* __restrict__ kcache,
    const _Float16* __restrict__ vcache, _Float16* __restrict__ out,
    const int* __restrict__ pos, int B, int H, int smax, float scale) {
  __shared__ float p_s[4096];
  int b = blockIdx.x / H, h = blockIdx.x % H;
  int lane = threadIdx.x;
  int hid = H * D;
  if (pos[b] < 0) return;  // idle slot: no scan, stale output row unused
  int n = pos[b] + 1;  // keys 0..pos[b] (this step's K already appended)

  // q for this head (fp16 registers: D=128 stays within budget)
  _Float16 q[D];
  {
    const _Float16* qrow = qkv + (int64_t)b * 3 * hid + h * D;
#pragma unroll
    for (int c = 0; c < D / 8; ++c)
      *(half8v*)(q + c * 8) = *(const half8v*)(qrow + c * 8);
  }
  const _Float16* K = kcache + ((int64_t)b * H + h) * smax * D;
  const _Float16* V = vcache + ((int64_t)b * H + h) * smax * D;

  // scores for this lane's keys
  float m = -3.0e38f;
  for (int t = lane; t < n; t += 64) {
    const _Float16* krow = K + (int64_t)t * D;
    float s = 0.f;
#pragma unroll
    for (int c = 0; c < D / 8; ++c) {
      half8v v = *(const half8v*)(krow + c * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        s += (float)q[c * 8 + j] * (float)((const _Float16*)&v)[j];
    }
    s *= scale;
    p_s[t] = s;
    m = fmaxf(m, s);
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
  __syncthreads();
  float l = 0.f;
  for (int t = lane; t < n; t += 64) {
    float e = __expf(p_s[t] - m);
    p_s[t] = e;
    l += e;
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) l += __shfl_xor(l, off, 64);
  __syncthreads();

  // PV: output elements d = lane (+64 for D=128)
  float acc[D / 64];
#pragma unroll
  for (int j = 0; j < D / 64; ++j) acc[j] = 0.f;
  for (int t = 0; t < n; ++t) {
    const _Float16* vrow = V + (int64_t)t * D;
    float p = p_s[t];
#pragma unroll
    for (int j = 0; j < D / 64; ++j)
      acc[j] += p * (float)vrow[j * 64 + lane];
  }
#pragma unroll
  for (int j = 0; j < D / 64; ++j)
    out[(int64_t)b * hid + h * D + j * 64 + lane] =
        (_Float16)(acc[j] / l);
}

void launch_decode_attention(const void* qkv, const void* kcache,
                             const void* vcache, void* out, const void* pos,
                             int B, int H, int smax, float scale,
                             hipStream_t stream, int D) {
  if (smax > 4096)
    throw std::runtime_error("decode_attention: smax > 4096 unsupported");
  if (D == 128)
    hipLaunchKernelGGL(decode_attention_kernel<128>, dim3(B * H), dim3(64),
                       0, stream, (const _Float16*)qkv,
                       (const _Float16*)kcache, (const _Float16*)vcache,
                       (_Float16*)out, (const int*)pos, B, H, smax, scale);
  else if (D == 64)
    hipLaunchKernelGGL(decode_attention_kernel<64>, dim3(B * H), dim3(64), 0,
                       stream, (const _Float16*)qkv, (const _Float16*)kcache,
                       (const _Float16*)vcache, (_Float16*)out,
                       (const int*)pos, B, H, smax, scale);
  else
    throw std::runtime_error("decode_attention: head_dim must be 64 or 128");
}

// ---- speculative-decoding verification chunk kernels ----
// The draft model proposes K tokens; the target verifies them in ONE
// chunked forward instead of K sequential replays. Rows are b-major then
// chunk position: row r = b*K + q corresponds to absolute position
// pos[b] + q.

// Scatter a chunk's K/V head rows into the caches at pos[b] + q.
__global__ __launch_bounds__(64) void kv_append_chunk_kernel(
    const _Float16* __restrict__ qkv, _Float16* __restrict__ kcache,
    _Float16* __restrict__ vcache, const int* __restrict__ pos, int B, int H,
    int K, int smax, int D) {
  int q = blockIdx.x % K;
  int bh = blockIdx.x / K;
  int b = bh / H, h = bh % H;
  int p0 = pos[b];
  if (p0 < 0) return;  // idle slot
  int p = p0 + q;
  if (p >= smax) p = smax - 1;
  int hid = H * D;
  for (int d = threadIdx.x; d < D; d += 64) {
    int64_t src = ((int64_t)b * K + q) * 3 * hid + h * D + d;
    int64_t dst = (((int64_t)b * H + h) * smax + p) * D + d;
    kcache[dst] = qkv[src + hid];
    vcache[dst] = qkv[src + 2 * hid];
  }
}

void launch_kv_append_chunk(const void* qkv, void* kcache, void* vcache,
                            const void* pos, int B, int H, int K, int smax,
                            hipStream_t stream, int D) {
  hipLaunchKernelGGL(kv_append_chunk_kernel, dim3(B * H * K), dim3(64), 0,
                     stream, (const _Float16*)qkv, (_Float16*)kcache,
                     (_Float16*)vcache, (const int*)pos, B, H, K, smax, D);
}

// Multi-query single-head attention against the cache: query row (b, q)
// attends keys 0 .. pos[b]+q (its own K already appended). One wave per
// (b, q, h) — the chunk's queries are causal within the chunk by
// construction of their lengths.
template <int D>  // head_dim 64 or 128
__global__ __launch_bounds__(64) void chunk_attention_kernel(
    const _Float16* __restrict__ qkv, const _Float16* __restrict__ kcache,
    const _Float16* __restrict__ vcache, _Float16* __restrict__ out,
    const int* __restrict__ pos, int B, int H, int K, int smax,
    float scale) {
  __shared__ float p_s[4096];
  int q = blockIdx.x % K;
  int bh = blockIdx.x / K;
  int b = bh / H, h = bh % H;
  int lane = threadIdx.x;
  int hid = H * D;
  int p0 = pos[b];
  if (p0 < 0) return;  // idle slot
  int n = p0 + q + 1;
  if (n > smax) n = smax;

  _Float16 qv[D];
  {
    const _Float16* qrow = qkv + ((int64_t)b * K + q) * 3 * hid + h * D;
#pragma unroll
    for (int c = 0; c < D / 8; ++c)
      *(half8v*)(qv + c * 8) = *(const half8v*)(qrow + c * 8);
  }
  const _Float16* Kc = kcache + ((int64_t)b * H + h) * smax * D;
  const _Float16* Vc = vcache + ((int64_t)b * H + h) * smax * D;

  float m = -3.0e38f;
  for (int t = lane; t < n; t += 64) {
    const _Float16* krow = Kc + (int64_t)t * D;
    float s = 0.f;
#pragma unroll
    for (int c = 0; c < D / 8; ++c) {
      half8v v = *(const half8v*)(krow + c * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        s += (float)qv[c * 8 + j] * (float)((const _Float16*)&v)[j];
    }
    s *= scale;
    p_s[t] = s;
    m = fmaxf(m, s);
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
  __syncthreads();
  float l = 0.f;
  for (int t = lane; t < n; t += 64) {
    float e = __expf(p_s[t] - m);
    p_s[t] = e;
    l += e;
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) l += __shfl_xor(l, off, 64);
  __syncthreads();
  float acc[D / 64];
#pragma unroll
  for (int j = 0; j < D / 64; ++j) acc[j] = 0.f;
  for (int t = 0; t < n; ++t) {
    const _Float16* vrow = Vc + (int64_t)t * D;
    float p = p_s[t];
#pragma unroll
    for (int j = 0; j < D / 64; ++j)
      acc[j] += p * (float)vrow[j * 64 + lane];
  }
#pragma unroll
  for (int j = 0; j < D / 64; ++j)
    out[((int64_t)b * K + q) * hid + h * D + j * 64 + lane] =
        (_Float16)(acc[j] / l);
}

void launch_chunk_attention(const void* qkv, const void* kcache,
                            const void* vcache, void* out, const void* pos,
                            int B, int H, int K, int smax, float scale,
                            hipStream_t stream, int D) {
  if (smax > 4096)
    throw std::runtime_error("chunk_attention: smax > 4096 unsupported");
  if (D == 128)
    hipLaunchKernelGGL(chunk_attention_kernel<128>, dim3(B * H * K),
                       dim3(64), 0, stream, (const _Float16*)qkv,
                       (const _Float16*)kcache, (const _Float16*)vcache,
                       (_Float16*)out, (const int*)pos, B, H, K, smax,
                       scale);
  else if (D == 64)
    hipLaunchKernelGGL(chunk_attention_kernel<64>, dim3(B * H * K), dim3(64),
                       0, stream, (const _Float16*)qkv,
                       (const _Float16*)kcache, (const _Float16*)vcache,
                       (_Float16*)out, (const int*)pos, B, H, K, smax,
                       scale);
  else
    throw std::runtime_error("chunk_attention: head_dim must be 64 or 128");
}

// Chunk embedding: out[b*K + q] = tok[ids[b*K + q]] + posemb[pos[b] + q].
__global__ void chunk_embed_kernel(const int* __restrict__ ids,
                                   const _Float16* __restrict__ tok,
                                   const _Float16* __restrict__ posemb,
                                   _Float16* __restrict__ out,
                                   const int* __restrict__ pos, int K,
                                   int smax, int hidden) {
  int q = blockIdx.x % K;
  int b = blockIdx.x / K;
  int p0 = pos[b];
  if (p0 < 0) return;
  int p = p0 + q;
  if (p >= smax) p = smax - 1;
  int64_t r = (int64_t)b * K + q;
  int64_t t = (int64_t)ids[r] * hidden;
  for (int i = threadIdx.x; i < hidden; i += blockDim.x)
    out[r * hidden + i] =
        (_Float16)((float)tok[t + i] +
                   (float)posemb[(int64_t)p * hidden + i]);
}

void launch_chunk_embed(const void* ids, const void* tok, const void* posemb,
                        void* out, const void* pos, int B, int K, int smax,
                        int hidden, hipStream_t stream) {
  hipLaunchKernelGGL(chunk_embed_kernel, dim3(B * K), dim3(256), 0, stream,
                     (const int*)ids, (const _Float16*)tok,
                     (const _Float16*)posemb, (_Float16*)out,
                     (const int*)pos, K, smax, hidden);
}

// Token + position embedding for one decode step: out[b] =
// tok[ids[b]] + posemb[pos]. ids are this step's B tokens.
__global__ void decode_embed_kernel(const int* __restrict__ ids,
                                    const _Float16* __restrict__ tok,
                                    const _Float16* __restrict__ posemb,
                                    _Float16* __restrict__ out,
                                    const int* __restrict__ pos, int hidden) {
  int b = blockIdx.x;
  int p = pos[b];
  if (p < 0) return;  // idle slot
  int64_t t = (int64_t)ids[b] * hidden;
  for (int i = threadIdx.x; i < hidden; i += blockDim.x)
    out[(int64_t)b * hidden + i] =
        (_Float16)((float)tok[t + i] + (float)posemb[(int64_t)p * hidden + i]);
}

void launch_decode_embed(const void* ids, const void* tok, const void* posemb,
                         void* out, const void* pos, int B, int hidden,
                         hipStream_t stream) {
  hipLaunchKernelGGL(decode_embed_kernel, dim3(B), dim3(256), 0, stream,
                     (const int*)ids, (const _Float16*)tok,
                     (const _Float16*)posemb, (_Float16*)out, (const int*)pos,
                     hidden);
}

// Advance every slot's position counter (last node of the captured
// decode step; clamped so replay past smax is safe).
__global__ void advance_pos_kernel(int* pos, int B, int smax) {
  int b = threadIdx.x;
  if (b < B && pos[b] >= 0) {  // idle slots stay parked at -1
    int p = pos[b] + 1;
    pos[b] = p >= smax ? smax - 1 : p;
  }
}

void launch_advance_pos(void* pos, int B, int smax, hipStream_t stream) {
  hipLaunchKernelGGL(advance_pos_kernel, dim3(1),
                     dim3((B + 63) / 64 * 64), 0, stream, (int*)pos, B,
                     smax);
}


// ---- fused decode GEMM (horizontal fusion for the latency-bound step) ----
// The decode step is kernel-COUNT bound (~4.6 us replay floor/kernel), so
// this kernel folds the surrounding elementwise work into the small-M
// (M = batch <= 64) GEMMs:
//   prologue PRO: 1 = LN(x), 2 = ADD_LN(x + r, also storing the new
//   residual stream h_out — blocks write identical values, benign),
//   3 = EMBED_LN(tok[ids[b]] + posemb[pos[b]])
//   epilogue EPI: 0 = +bias, 1 = gelu(+bias), 2 = +bias AND scatter the
//   K/V column ranges into the caches at pos[b] (replaces kv_append)
// One block per 64 output columns; rows are the whole batch. LN stats are
// computed in-block (4 lanes per row, shfl-combined) and the normalized
// A-tile is ds_written per K-step with the same XOR swizzle the MFMA
// fragment reads expect. Cuts the GPT-2 step from ~8 to 5 kernels/layer.
template <int PRO, int EPI>
__global__ __launch_bounds__(256) void decode_gemm_fused_kernel(
    const _Float16* __restrict__ x, const _Float16* __restrict__ r,
    _Float16* __restrict__ h_out, const float* __restrict__ gamma,
    const float* __restrict__ beta, const _Float16* __restrict__ Bw,
    const float* __restrict__ bias, _Float16* __restrict__ C,
    const int* __restrict__ ids, const _Float16* __restrict__ tok,
    const _Float16* __restrict__ posemb, const int* __restrict__ pos,
    _Float16* __restrict__ kcache, _Float16* __restrict__ vcache, int M,
    int N, int K, int heads, int smax, float eps) {
  constexpr int BM = 64, BN = 64;
  __shared__ __attribute__((aligned(16))) char smem[2 * (BM + BN) * 128];
  __shared__ float sm_mean[BM], sm_inv[BM];
  // gamma/beta staged once per block: the per-tile normalize would
  // otherwise expose a global-latency chain every iteration
  __shared__ float sm_gamma[2048], sm_beta[2048];
  uint32_t lds0 = (uint32_t)(uintptr_t)&smem[0];
  constexpr int kABytes = BM * 128;
  constexpr int kBuf = (BM + BN) * 128;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1;
  const int wc = wave & 1;
  const int n0 = (int)blockIdx.x * BN;
  int cur_kbase = 0;  // K-base of the tile currently being staged

  // source value of the pre-norm row (x + r / embed), 8 elems at kk
  auto load8 = [&](int row, int kk, _Float16* v8) {
    if (PRO == 3) {
      int p = pos[row];
      if (p < 0) p = 0;
      const _Float16* trow = tok + (int64_t)ids[row] * K + kk;
      const _Float16* prow = posemb + (int64_t)p * K + kk;
      half8v a = *(const half8v*)trow;
      half8v b = *(const half8v*)prow;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        v8[j] = (_Float16)((float)((const _Float16*)&a)[j] +
                           (float)((const _Float16*)&b)[j]);
    } else if (PRO == 2) {
      half8v a = *(const half8v*)(x + (int64_t)row * K + kk);
      half8v b = *(const half8v*)(r + (int64_t)row * K + kk);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        v8[j] = (_Float16)((float)((const _Float16*)&a)[j] +
                           (float)((const _Float16*)&b)[j]);
    } else {
      *(half8v*)v8 = *(const half8v*)(x + (int64_t)row * K + kk);
    }
  };

  // ---- stage gamma/beta + LN stats ----
  for (int i = tid; i < K; i += 256) {
    sm_gamma[i] = gamma ? gamma[i] : 1.0f;
    sm_beta[i] = beta ? beta[i] : 0.0f;
  }
  {
    int row = tid >> 2;
    int part = tid & 3;
    float s = 0.f, ss = 0.f;
    if (row < M) {
      for (int kk = part * 8; kk < K; kk += 32) {
        _Float16 v8[8];
        load8(row, kk, v8);
        // block 0 persists the new residual stream h_out = x + r. h_out
        // MUST be a different buffer than r (ping-pong in the session):
        // other blocks re-read r during their own staging, so an in-place
        // update would race. Every block computes identical values; the
        // n0 gate just avoids duplicate HBM traffic.
        if ((PRO == 2 || PRO == 3) && n0 == 0 && h_out)
          *(half8v*)(h_out + (int64_t)row * K + kk) = *(const half8v*)v8;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float f = (float)v8[j];
          s += f;
          ss += f * f;
        }
      }
    }
    // combine the 4 partial lanes of this row (consecutive lanes)
    s += __shfl_xor(s, 1, 64);
    s += __shfl_xor(s, 2, 64);
    ss += __shfl_xor(ss, 1, 64);
    ss += __shfl_xor(ss, 2, 64);
    if (part == 0 && row < M) {
      float mean = s / K;
      float var = ss / K - mean * mean;
      sm_mean[row] = mean;
      sm_inv[row] = rsqrtf(var + eps);
    }
  }
  __syncthreads();

  typename Mfma16x16x32<_Float16>::accv acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0, 0, 0, 0};

  const int ktiles = K / 64;
  // Register-prefetch pipeline (the attention kernel's proven pattern):
  // A and B tiles are prefetched into REGISTERS (not glds), so
  // __syncthreads() between ds_writes and the MFMAs only waits lgkmcnt —
  // the next tile's global loads stay in flight under the compute, and
  // the ds_writes' register dependencies are the only load waits.
  _Float16 a_reg[2][8];
  _Float16 b_reg[2][8];
  auto prefetch = [&](int t) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = c * 256 + tid;      // 0..511
      int row = idx >> 3;           // 0..63
      int cb = (idx & 7) * 16;
      int kk = t * 64 + cb / 2;
      int ar = row < M ? row : M - 1;
      load8(ar, kk, a_reg[c]);
      int br = n0 + row;
      if (br >= N) br = N - 1;
      *(half8v*)b_reg[c] = *(const half8v*)(Bw + (int64_t)br * K + kk);
    }
  };
  auto stage_regs = [&](char* abase) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int idx = c * 256 + tid;
      int row = idx >> 3;
      int cb = (idx & 7) * 16;
      int kk = (cb / 2);  // column within the tile for gamma/beta lookup
      (void)kk;
      float mean = sm_mean[row < M ? row : M - 1];
      float inv = sm_inv[row < M ? row : M - 1];
      _Float16 o8[8];
      int kcol = 0;  // filled below per element
      (void)kcol;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int kg = cur_kbase + cb / 2 + j;
        o8[j] = (_Float16)(((float)a_reg[c][j] - mean) * inv * sm_gamma[kg] +
                           sm_beta[kg]);
      }
      *(short8v*)(abase + row * 128 + (cb ^ ((row & 7) << 4))) =
          *(const short8v*)o8;
      *(short8v*)(abase + kABytes + row * 128 + (cb ^ ((row & 7) << 4))) =
          *(const short8v*)b_reg[c];
    }
  };

  char* abase = smem;  // single slot: barrier-write-barrier-compute
  cur_kbase = 0;
  prefetch(0);
  for (int t = 0; t < ktiles; ++t) {
    __syncthreads();   // prior tile's LDS reads complete before overwrite
    cur_kbase = t * 64;
    stage_regs(abase);             // waits only the prefetched registers
    if (t + 1 < ktiles) prefetch(t + 1);  // overlaps the MFMAs below
    __syncthreads();               // ds_writes visible (lgkmcnt)
    mfma_tile<_Float16, BM, BN>(abase, abase + kABytes, lane, wr, wc, acc);
  }

  // ---- epilogue: bias (+gelu) (+K/V scatter) ----
  const int hid = (EPI == 2) ? N / 3 : 0;  // qkv gemm: N = 3*H*64
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int col = n0 + wc * 32 + j * 16 + (lane & 15);
      if (col >= N) continue;
      float bi = bias ? bias[col] : 0.0f;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int row = wr * 32 + i * 16 + ((lane >> 4) << 2) + rr;
        if (row >= M) continue;
        float v = (float)acc[i][j][rr] + bi;
        if (EPI == 1) v = gelu_tanh(v);
        _Float16 hv = (_Float16)v;
        C[(int64_t)row * N + col] = hv;
        if (EPI == 2) {
          int p = pos[row];
          if (p >= 0 && col >= hid) {  // K or V column range
            if (p >= smax) p = smax - 1;
            int rel = col - hid;
            bool is_v = rel >= hid;
            if (is_v) rel -= hid;
            int h_idx = rel >> 6, d = rel & 63;
            _Float16* cache = is_v ? vcache : kcache;
            cache[(((int64_t)row * heads + h_idx) * smax + p) * 64 + d] = hv;
          }
        }
      }
    }
  }
}

void launch_decode_gemm_fused(int pro, int epi, const void* x, const void* r,
                              void* h_out, const float* gamma,
                              const float* beta, const void* Bw,
                              const float* bias, void* C, const void* ids,
                              const void* tok, const void* posemb,
                              const void* pos, void* kcache, void* vcache,
                              int M, int N, int K, int heads, int smax,
                              float eps, hipStream_t stream) {
  if (M > 64) throw std::runtime_error("decode_gemm_fused: M > 64");
  if (K % 64 != 0) throw std::runtime_error("decode_gemm_fused: K % 64");
  if (K > 2048) throw std::runtime_error("decode_gemm_fused: K > 2048");
  dim3 grid((unsigned)cdiv(N, 64));
  dim3 block(256);
  auto L = [&](auto pro_c, auto epi_c) {
    hipLaunchKernelGGL(
        (decode_gemm_fused_kernel<decltype(pro_c)::value,
                                  decltype(epi_c)::value>),
        grid, block, 0, stream, (const _Float16*)x, (const _Float16*)r,
        (_Float16*)h_out, gamma, beta, (const _Float16*)Bw, bias,
        (_Float16*)C, (const int*)ids, (const _Float16*)tok,
        (const _Float16*)posemb, (const int*)pos, (_Float16*)kcache,
        (_Float16*)vcache, M, N, K, heads, smax, eps);
  };
  using I1 = std::integral_constant<int, 1>;
  using I2 = std::integral_constant<int, 2>;
  using I3 = std::integral_constant<int, 3>;
  using E0 = std::integral_constant<int, 0>;
  using E1 = std::integral_constant<int, 1>;
  using E2 = std::integral_constant<int, 2>;
  if (pro == 1 && epi == 0) L(I1{}, E0{});
  else if (pro == 1 && epi == 1) L(I1{}, E1{});
  else if (pro == 1 && epi == 2) L(I1{}, E2{});
  else if (pro == 2 && epi == 0) L(I2{}, E0{});
  else if (pro == 2 && epi == 1) L(I2{}, E1{});
  else if (pro == 2 && epi == 2) L(I2{}, E2{});
  else if (pro == 3 && epi == 2) L(I3{}, E2{});
  else if (pro == 3 && epi == 0) L(I3{}, E0{});
  else throw std::runtime_error("decode_gemm_fused: bad pro/epi combo");
}


// ---- paged KV cache (vLLM-style block tables) ----
// Many concurrent sessions share ONE fixed physical pool per layer; each
// slot maps logical 64-position pages to physical page ids through a
// device block table [B][max_pages] (filled by the host as positions
// grow; reset/idle returns pages to the host free list, so parked slots
// hold no memory). Page layout: [page][64 positions][H][64] — a
// position's per-head row stays one contiguous 128-B cacheline.
__global__ __launch_bounds__(64) void kv_append_paged_kernel(
    const _Float16* __restrict__ qkv, _Float16* __restrict__ kpool,
    _Float16* __restrict__ vpool, const int* __restrict__ table,
    const int* __restrict__ pos, int B, int H, int max_pages, int D) {
  int b = blockIdx.x / H, h = blockIdx.x % H;
  int p = pos[b];
  if (p < 0) return;  // idle slot
  int page = table[b * max_pages + (p >> 6)];
  if (page < 0) return;  // unmapped (host error) — fail soft, not wild
  int hid = H * D;
  for (int d = threadIdx.x; d < D; d += 64) {
    int64_t src = (int64_t)b * 3 * hid + h * D + d;
    int64_t dst = (((int64_t)page * 64 + (p & 63)) * H + h) * D + d;
    kpool[dst] = qkv[src + hid];
    vpool[dst] = qkv[src + 2 * hid];
  }
}

// ---- paged variants of the chunk / prefill cache writers + reader ----
// (speculative verification and fused prefill for paged sessions: same
// math as the dense chunk kernels, addresses resolved through the
// per-slot page table; pages host-mapped before launch.)
__global__ __launch_bounds__(64) void kv_append_chunk_paged_kernel(
    const _Float16* __restrict__ qkv, _Float16* __restrict__ kpool,
    _Float16* __restrict__ vpool, const int* __restrict__ table,
    const int* __restrict__ pos, int B, int H, int K, int max_pages,
    int D) {
  int q = blockIdx.x % K;
  int bh = blockIdx.x / K;
  int b = bh / H, h = bh % H;
  int p0 = pos[b];
  if (p0 < 0) return;  // idle slot
  int p = p0 + q;
  int page = table[b * max_pages + (p >> 6)];
  if (page < 0) return;  // unmapped (host error) — fail soft
  int hid = H * D;
  for (int d = threadIdx.x; d < D; d += 64) {
    int64_t src = ((int64_t)b * K + q) * 3 * hid + h * D + d;
    int64_t dst = (((int64_t)page * 64 + (p & 63)) * H + h) * D + d;
    kpool[dst] = qkv[src + hid];
    vpool[dst] = qkv[src + 2 * hid];
  }
}

void launch_kv_append_chunk_paged(const void* qkv, void* kpool, void* vpool,
                                  const void* table, const void* pos, int B,
                                  int H, int K, int max_pages,
                                  hipStream_t stream, int D) {
  hipLaunchKernelGGL(kv_append_chunk_paged_kernel, dim3(B * H * K),
                     dim3(64), 0, stream, (const _Float16*)qkv,
                     (_Float16*)kpool, (_Float16*)vpool, (const int*)table,
                     (const int*)pos, B, H, K, max_pages, D);
}

__global__ __launch_bounds__(64) void kv_append_range_paged_kernel(
    const _Float16* __restrict__ qkv, _Float16* __restrict__ kpool,
    _Float16* __restrict__ vpool, const int* __restrict__ table, int B,
    int H, int P, int max_pages, int D) {
  int p = blockIdx.x % P;
  int bh = blockIdx.x / P;
  int b = bh / H, h = bh % H;
  int page = table[b * max_pages + (p >> 6)];
  if (page < 0) return;
  int hid = H * D;
  for (int d = threadIdx.x; d < D; d += 64) {
    int64_t src = ((int64_t)b * P + p) * 3 * hid + h * D + d;
    int64_t dst = (((int64_t)page * 64 + (p & 63)) * H + h) * D + d;
    kpool[dst] = qkv[src + hid];
    vpool[dst] = qkv[src + 2 * hid];
  }
}

void launch_kv_append_range_paged(const void* qkv, void* kpool, void* vpool,
                                  const void* table, int B, int H, int P,
                                  int max_pages, hipStream_t stream, int D) {
  hipLaunchKernelGGL(kv_append_range_paged_kernel, dim3(B * H * P),
                     dim3(64), 0, stream, (const _Float16*)qkv,
                     (_Float16*)kpool, (_Float16*)vpool, (const int*)table,
                     B, H, P, max_pages, D);
}

template <int D>  // head_dim 64 or 128
__global__ __launch_bounds__(64) void chunk_attention_paged_kernel(
    const _Float16* __restrict__ qkv, const _Float16* __restrict__ kpool,
    const _Float16* __restrict__ vpool, _Float16* __restrict__ out,
    const int* __restrict__ table, const int* __restrict__ pos, int B,
    int H, int K, int max_pages, float scale) {
  __shared__ float p_s[4096];
  int q = blockIdx.x % K;
  int bh = blockIdx.x / K;
  int b = bh / H, h = bh % H;
  int lane = threadIdx.x;
  int hid = H * D;
  int p0 = pos[b];
  if (p0 < 0) return;  // idle slot
  int n = p0 + q + 1;

  _Float16 qv[D];
  {
    const _Float16* qrow = qkv + ((int64_t)b * K + q) * 3 * hid + h * D;
#pragma unroll
    for (int c = 0; c < D / 8; ++c)
      *(half8v*)(qv + c * 8) = *(const half8v*)(qrow + c * 8);
  }
  const int* trow = table + b * max_pages;

  float m = -3.0e38f;
  for (int t = lane; t < n; t += 64) {
    int page = trow[t >> 6];
    const _Float16* krow =
        kpool + (((int64_t)page * 64 + (t & 63)) * H + h) * D;
    float sc = 0.f;
#pragma unroll
    for (int c = 0; c < D / 8; ++c) {
      half8v v = *(const half8v*)(krow + c * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        sc += (float)qv[c * 8 + j] * (float)((const _Float16*)&v)[j];
    }
    sc *= scale;
    p_s[t] = sc;
    m = fmaxf(m, sc);
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
  __syncthreads();
  float l = 0.f;
  for (int t = lane; t < n; t += 64) {
    float e = __expf(p_s[t] - m);
    p_s[t] = e;
    l += e;
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) l += __shfl_xor(l, off, 64);
  __syncthreads();
  float acc[D / 64];
#pragma unroll
  for (int j = 0; j < D / 64; ++j) acc[j] = 0.f;
  for (int t = 0; t < n; ++t) {
    int page = trow[t >> 6];
    const _Float16* vrow =
        vpool + (((int64_t)page * 64 + (t & 63)) * H + h) * D;
    float pp = p_s[t];
#pragma unroll
    for (int j = 0; j < D / 64; ++j)
      acc[j] += pp * (float)vrow[j * 64 + lane];
  }
#pragma unroll
  for (int j = 0; j < D / 64; ++j)
    out[((int64_t)b * K + q) * hid + h * D + j * 64 + lane] =
        (_Float16)(acc[j] / l);
}

void launch_chunk_attention_paged(const void* qkv, const void* kpool,
                                  const void* vpool, void* out,
                                  const void* table, const void* pos, int B,
                                  int H, int K, int max_pages, float scale,
                                  hipStream_t stream, int D) {
  if (D == 128)
    hipLaunchKernelGGL(chunk_attention_paged_kernel<128>, dim3(B * H * K),
                       dim3(64), 0, stream, (const _Float16*)qkv,
                       (const _Float16*)kpool, (const _Float16*)vpool,
                       (_Float16*)out, (const int*)table, (const int*)pos,
                       B, H, K, max_pages, scale);
  else if (D == 64)
    hipLaunchKernelGGL(chunk_attention_paged_kernel<64>, dim3(B * H * K),
                       dim3(64), 0, stream, (const _Float16*)qkv,
                       (const _Float16*)kpool, (const _Float16*)vpool,
                       (_Float16*)out, (const int*)table, (const int*)pos,
                       B, H, K, max_pages, scale);
  else
    throw std::runtime_error("chunk_attention_paged: head_dim 64/128");
}

void launch_kv_append_paged(const void* qkv, void* kpool, void* vpool,
                            const void* table, const void* pos, int B, int H,
                            int max_pages, hipStream_t stream, int D) {
  hipLaunchKernelGGL(kv_append_paged_kernel, dim3(B * H), dim3(64), 0,
                     stream, (const _Float16*)qkv, (_Float16*)kpool,
                     (_Float16*)vpool, (const int*)table, (const int*)pos,
                     B, H, max_pages, D);
}

template <int D>  // head_dim 64 or 128
__global__ __launch_bounds__(64) void decode_attention_paged_kernel(
    const _Float16* __restrict__ qkv, const _Float16* __restrict__ kpool,
    const _Float16* __restrict__ vpool, _Float16* __restrict__ out,
    const int* __restrict__ table, const int* __restrict__ pos, int B, int H,
    int max_pages, float scale) {
  __shared__ float p_s[4096];
  int b = blockIdx.x / H, h = blockIdx.x % H;
  int lane = threadIdx.x;
  int hid = H * D;
  if (pos[b] < 0) return;  // idle slot
  int n = pos[b] + 1;

  _Float16 q[D];
  {
    const _Float16* qrow = qkv + (int64_t)b * 3 * hid + h * D;
#pragma unroll
    for (int c = 0; c < D / 8; ++c)
      *(half8v*)(q + c * 8) = *(const half8v*)(qrow + c * 8);
  }
  const int* tab = table + (int64_t)b * max_pages;

  float m = -3.0e38f;
  for (int t = lane; t < n; t += 64) {
    int64_t rowoff = (((int64_t)tab[t >> 6] * 64 + (t & 63)) * H + h) * D;
    const _Float16* krow = kpool + rowoff;
    float s = 0.f;
#pragma unroll
    for (int c = 0; c < D / 8; ++c) {
      half8v v = *(const half8v*)(krow + c * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        s += (float)q[c * 8 + j] * (float)((const _Float16*)&v)[j];
    }
    s *= scale;
    p_s[t] = s;
    m = fmaxf(m, s);
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) m = fmaxf(m, __shfl_xor(m, off, 64));
  __syncthreads();
  float l = 0.f;
  for (int t = lane; t < n; t += 64) {
    float e = __expf(p_s[t] - m);
    p_s[t] = e;
    l += e;
  }
#pragma unroll
  for (int off = 32; off; off >>= 1) l += __shfl_xor(l, off, 64);
  __syncthreads();
  float acc[D / 64];
#pragma unroll
  for (int j = 0; j < D / 64; ++j) acc[j] = 0.f;
  for (int t = 0; t < n; ++t) {
    int64_t rowoff = (((int64_t)tab[t >> 6] * 64 + (t & 63)) * H + h) * D;
    float p = p_s[t];
#pragma unroll
    for (int j = 0; j < D / 64; ++j)
      acc[j] += p * (float)vpool[rowoff + j * 64 + lane];
  }
#pragma unroll
  for (int j = 0; j < D / 64; ++j)
    out[(int64_t)b * hid + h * D + j * 64 + lane] = (_Float16)(acc[j] / l);
}

void launch_decode_attention_paged(const void* qkv, const void* kpool,
                                   const void* vpool, void* out,
                                   const void* table, const void* pos, int B,
                                   int H, int max_pages, float scale,
                                   hipStream_t stream, int D) {
  if (D == 128)
    hipLaunchKernelGGL(decode_attention_paged_kernel<128>, dim3(B * H),
                       dim3(64), 0, stream, (const _Float16*)qkv,
                       (const _Float16*)kpool, (const _Float16*)vpool,
                       (_Float16*)out, (const int*)table, (const int*)pos,
                       B, H, max_pages, scale);
  else if (D == 64)
    hipLaunchKernelGGL(decode_attention_paged_kernel<64>, dim3(B * H),
                       dim3(64), 0, stream, (const _Float16*)qkv,
                       (const _Float16*)kpool, (const _Float16*)vpool,
                       (_Float16*)out, (const int*)table, (const int*)pos,
                       B, H, max_pages, scale);
  else
    throw std::runtime_error(
        "decode_attention_paged: head_dim must be 64 or 128");
}

}  // namespace trtlab
