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
    int smax) {
  int b = blockIdx.x / H, h = blockIdx.x % H;
  int d = threadIdx.x;  // 0..63
  int p = pos[b];
  if (p < 0) return;  // idle slot
  int hid = H * 64;
  int64_t src = (int64_t)b * 3 * hid + h * 64 + d;
  int64_t dst = (((int64_t)b * H + h) * smax + p) * 64 + d;
  kcache[dst] = qkv[src + hid];
  vcache[dst] = qkv[src + 2 * hid];
}

void launch_kv_append(const void* qkv, void* kcache, void* vcache,
                      const void* pos, int B, int H, int smax,
                      hipStream_t stream) {
  hipLaunchKernelGGL(kv_append_kernel, dim3(B * H), dim3(64), 0, stream,
                     (const _Float16*)qkv, (_Float16*)kcache,
                     (_Float16*)vcache, (const int*)pos, B, H, smax);
}

// Batch-scatter a full prompt's K/V head rows (qkv [B*P, 3*H*64], rows
// ordered b-major then position) into the caches at positions [0, P).
// Grid B*H*P; fused prefill runs the prompt through the full-sequence
// kernels once, then decode continues from position P.
__global__ __launch_bounds__(64) void kv_append_range_kernel(
    const _Float16* __restrict__ qkv, _Float16* __restrict__ kcache,
    _Float16* __restrict__ vcache, int B, int H, int P, int smax) {
  int p = blockIdx.x % P;
  int bh = blockIdx.x / P;
  int b = bh / H, h = bh % H;
  int d = threadIdx.x;
  int hid = H * 64;
  int64_t src = ((int64_t)b * P + p) * 3 * hid + h * 64 + d;
  int64_t dst = (((int64_t)b * H + h) * smax + p) * 64 + d;
  kcache[dst] = qkv[src + hid];
  vcache[dst] = qkv[src + 2 * hid];
}

void launch_kv_append_range(const void* qkv, void* kcache, void* vcache,
                            int B, int H, int P, int smax,
                            hipStream_t stream) {
  hipLaunchKernelGGL(kv_append_range_kernel, dim3(B * H * P), dim3(64), 0,
                     stream, (const _Float16*)qkv, (_Float16*)kcache,
                     (_Float16*)vcache, B, H, P, smax);
}

// Single-query attention against the cache: out[b, h*64+d] =
// softmax(q . K[0..pos]) @ V[0..pos]. One wave per (b, h); each lane owns
// keys lane, lane+64, ... for the score pass (its K rows are whole
// 128-byte cachelines), then output element d = lane for the PV pass with
// the probabilities broadcast through LDS. Smax <= 4096.
__global__ __launch_bounds__(64) void decode_attention_kernel(
    const _Float16* __restrict__ qkv, const _Float16* __restrict__ kcache,
    const _Float16* __restrict__ vcache, _Float16* __restrict__ out,
    const int* __restrict__ pos, int B, int H, int smax, float scale) {
  __shared__ float p_s[4096];
  int b = blockIdx.x / H, h = blockIdx.x % H;
  int lane = threadIdx.x;
  int hid = H * 64;
  if (pos[b] < 0) return;  // idle slot: no scan, stale output row unused
  int n = pos[b] + 1;  // keys 0..pos[b] (this step's K already appended)

  // q for this head, one element per lane
  float q[64];
  {
    const _Float16* qrow = qkv + (int64_t)b * 3 * hid + h * 64;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      half8v v = *(const half8v*)(qrow + c * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) q[c * 8 + j] = (float)((const _Float16*)&v)[j];
    }
  }
  const _Float16* K = kcache + ((int64_t)b * H + h) * smax * 64;
  const _Float16* V = vcache + ((int64_t)b * H + h) * smax * 64;

  // scores for this lane's keys
  float m = -3.0e38f;
  for (int t = lane; t < n; t += 64) {
    const _Float16* krow = K + (int64_t)t * 64;
    float s = 0.f;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      half8v v = *(const half8v*)(krow + c * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        s += q[c * 8 + j] * (float)((const _Float16*)&v)[j];
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

  // PV: output element d = lane
  float acc = 0.f;
  for (int t = 0; t < n; ++t) acc += p_s[t] * (float)V[(int64_t)t * 64 + lane];
  out[(int64_t)b * hid + h * 64 + lane] = (_Float16)(acc / l);
}

void launch_decode_attention(const void* qkv, const void* kcache,
                             const void* vcache, void* out, const void* pos,
                             int B, int H, int smax, float scale,
                             hipStream_t stream) {
  if (smax > 4096)
    throw std::runtime_error("decode_attention: smax > 4096 unsupported");
  hipLaunchKernelGGL(decode_attention_kernel, dim3(B * H), dim3(64), 0,
                     stream, (const _Float16*)qkv, (const _Float16*)kcache,
                     (const _Float16*)vcache, (_Float16*)out, (const int*)pos,
                     B, H, smax, scale);
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
    int K, int smax) {
  int q = blockIdx.x % K;
  int bh = blockIdx.x / K;
  int b = bh / H, h = bh % H;
  int d = threadIdx.x;
  int p0 = pos[b];
  if (p0 < 0) return;  // idle slot
  int p = p0 + q;
  if (p >= smax) p = smax - 1;
  int hid = H * 64;
  int64_t src = ((int64_t)b * K + q) * 3 * hid + h * 64 + d;
  int64_t dst = (((int64_t)b * H + h) * smax + p) * 64 + d;
  kcache[dst] = qkv[src + hid];
  vcache[dst] = qkv[src + 2 * hid];
}

void launch_kv_append_chunk(const void* qkv, void* kcache, void* vcache,
                            const void* pos, int B, int H, int K, int smax,
                            hipStream_t stream) {
  hipLaunchKernelGGL(kv_append_chunk_kernel, dim3(B * H * K), dim3(64), 0,
                     stream, (const _Float16*)qkv, (_Float16*)kcache,
                     (_Float16*)vcache, (const int*)pos, B, H, K, smax);
}

// Multi-query single-head attention against the cache: query row (b, q)
// attends keys 0 .. pos[b]+q (its own K already appended). One wave per
// (b, q, h) — the chunk's queries are causal within the chunk by
// construction of their lengths.
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
  int hid = H * 64;
  int p0 = pos[b];
  if (p0 < 0) return;  // idle slot
  int n = p0 + q + 1;
  if (n > smax) n = smax;

  float qv[64];
  {
    const _Float16* qrow =
        qkv + ((int64_t)b * K + q) * 3 * hid + h * 64;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      half8v v = *(const half8v*)(qrow + c * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        qv[c * 8 + j] = (float)((const _Float16*)&v)[j];
    }
  }
  const _Float16* Kc = kcache + ((int64_t)b * H + h) * smax * 64;
  const _Float16* Vc = vcache + ((int64_t)b * H + h) * smax * 64;

  float m = -3.0e38f;
  for (int t = lane; t < n; t += 64) {
    const _Float16* krow = Kc + (int64_t)t * 64;
    float s = 0.f;
#pragma unroll
    for (int c = 0; c < 8; ++c) {
      half8v v = *(const half8v*)(krow + c * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        s += qv[c * 8 + j] * (float)((const _Float16*)&v)[j];
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
  float acc = 0.f;
  for (int t = 0; t < n; ++t)
    acc += p_s[t] * (float)Vc[(int64_t)t * 64 + lane];
  out[((int64_t)b * K + q) * hid + h * 64 + lane] = (_Float16)(acc / l);
}

void launch_chunk_attention(const void* qkv, const void* kcache,
                            const void* vcache, void* out, const void* pos,
                            int B, int H, int K, int smax, float scale,
                            hipStream_t stream) {
  if (smax > 4096)
    throw std::runtime_error("chunk_attention: smax > 4096 unsupported");
  hipLaunchKernelGGL(chunk_attention_kernel, dim3(B * H * K), dim3(64), 0,
                     stream, (const _Float16*)qkv, (const _Float16*)kcache,
                     (const _Float16*)vcache, (_Float16*)out,
                     (const int*)pos, B, H, K, smax, scale);
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

}  // namespace trtlab
