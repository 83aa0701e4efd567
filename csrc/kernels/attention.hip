// trtlab_amd — fused multi-head self-attention for the BERT encoder path
// (gfx950). SURVEY.md §2.8 item 3: batched GEMM + softmax fused in one
// kernel for seq=128.
//
// Input: qkv [B*S, 3*H*D] (the fused QKV projection output, column blocks
// q|k|v, each H*D with head-major (h,d) minor d). Output: [B*S, H*D].
//
// One workgroup per (b, h): Q,K,V head tiles live entirely in LDS
// (S=128, D=64 -> 16 KiB each). 4 waves; each wave owns 32 query rows:
//   QK^T via mfma_f32_16x16x32_f16 (K tile is already "bt" layout),
//   row softmax fully wave-local (rows live on 16-lane groups; shfl_xor),
//   P staged to per-wave LDS as two [32][64] tiles, PV against V^T staged
//   transposed at load. fp32 accumulation throughout; scores scaled by
//   1/sqrt(D). No attention mask (synthetic full-length sequences; masked
//   variant planned).
// Constraints: S == 128, D == 64.
#include "gemm_common.h"

namespace trtlab {

template <typename T>
__global__ __launch_bounds__(256) void attention_kernel(
    const T* __restrict__ qkv, T* __restrict__ out, int B, int S, int H,
    int D, float scale) {
  // LDS: Q [128][64] | K [128][64] | Vt [64][128->2x[64][64]] | P 4x[2x[32][64]]
  __shared__ __attribute__((aligned(16))) char smem[16384 * 3 + 4 * 8192];
  char* Qs = smem;
  char* Ks = smem + 16384;
  char* Vt = smem + 32768;          // two [64][64] tiles: kt*8192
  char* Ps = smem + 49152;          // per wave: wave*8192, two [32][64] tiles

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int bh = blockIdx.x;
  const int b = bh / H;
  const int h = bh % H;
  const int hid = H * D;           // 768 for BERT-base
  const int row_stride = 3 * hid;  // qkv row stride in elements

  const T* base = qkv + (int64_t)b * S * row_stride;
  const int qoff = h * D;
  const int koff = hid + h * D;
  const int voff = 2 * hid + h * D;

  auto swz = [](uint32_t row, uint32_t colbyte) {
    return row * 128 + (colbyte ^ ((row & 7) << 4));
  };

  // ---- stage Q, K (swizzled 16-B chunks), Vt (transposed scatter) ----
  // Q/K: 128 rows x 128 B/row -> 1024 16-B chunks each, 4 per thread.
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    int idx = c * 256 + tid;       // 0..1023
    int row = idx >> 3;            // 0..127
    int cb = (idx & 7) * 16;       // byte offset within 128-B row
    const T* src = base + (int64_t)row * row_stride + qoff + cb / 2;
    *(short8v*)(Qs + swz(row, cb)) = *(const short8v*)src;
    const T* ksrc = base + (int64_t)row * row_stride + koff + cb / 2;
    *(short8v*)(Ks + swz(row, cb)) = *(const short8v*)ksrc;
  }
  // Vt: read v[key][dd..dd+8] (16 B), scatter transposed to Vt[kt][d][key%64].
  // 128 keys x 8 chunks = 1024 chunks, 4 per thread.
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    int ci = c * 256 + tid;        // 0..1023
    int key = ci >> 3;
    int dd = (ci & 7) * 8;
    const T* vsrc = base + (int64_t)key * row_stride + voff + dd;
    short8v v = *(const short8v*)vsrc;
    char* tile = Vt + (key >> 6) * 8192;
    int kcol = key & 63;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      *(T*)(tile + swz(dd + j, kcol * 2)) = ((const T*)&v)[j];
    }
  }
  __syncthreads();

  // ---- QK^T for this wave's 32 query rows ----
  using MF = Mfma16x16x32<T>;
  const int qrow0 = wave * 32;
  f32x4 sacc[2][8];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) sacc[i][j] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int ks = 0; ks < 2; ++ks) {
    uint32_t kbyte = ks * 64 + ((lane >> 4) << 4);
    typename MF::frag qf[2], kf[8];
#pragma unroll
    for (int f = 0; f < 2; ++f)
      qf[f] = *(const typename MF::frag*)(Qs +
               swz(qrow0 + f * 16 + (lane & 15), kbyte));
#pragma unroll
    for (int f = 0; f < 8; ++f)
      kf[f] = *(const typename MF::frag*)(Ks + swz(f * 16 + (lane & 15), kbyte));
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < 8; ++j)
        sacc[i][j] = MF::run(qf[i], kf[j], sacc[i][j]);
  }

  // ---- row softmax (rows live on 16-lane groups: shfl_xor 1,2,4,8) ----
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m = -3.0e38f;
#pragma unroll
      for (int j = 0; j < 8; ++j) m = fmaxf(m, sacc[i][j][r] * scale);
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        m = fmaxf(m, __shfl_xor(m, off, 64));
      float s = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float e = __expf(sacc[i][j][r] * scale - m);
        sacc[i][j][r] = e;
        s += e;
      }
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) s += __shfl_xor(s, off, 64);
      float inv = 1.0f / s;
#pragma unroll
      for (int j = 0; j < 8; ++j) sacc[i][j][r] *= inv;
    }
  }

  // ---- P -> per-wave LDS as two [32][64] fp16 tiles ----
  char* P = Ps + wave * 8192;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int colg = j * 16 + (lane & 15);
      char* tile = P + (colg >> 6) * 4096;
      int col = colg & 63;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = i * 16 + ((lane >> 4) << 2) + r;
        *(T*)(tile + ((uint32_t)row * 128 + ((col * 2) ^ ((row & 7) << 4)))) =
            (T)sacc[i][j][r];
      }
    }
  }
  // Force completion of the P ds_writes before the PV ds_reads. A bare
  // same-wave write->read *should* be ordered by the compiler's lgkmcnt
  // bookkeeping, but was observed to return stale P for some blocks on
  // real data (one (b,h) block NaN per launch); the barrier makes the
  // ordering explicit and costs nothing at this kernel's size.
  __syncthreads();

  // ---- PV: out_tile[32 rows][64 d] = P[32][128] @ Vt^T ----
  f32x4 oacc[2][4];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) oacc[i][j] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int kt = 0; kt < 2; ++kt) {
    const char* Pt = P + kt * 4096;
    const char* Vk = Vt + kt * 8192;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      uint32_t kbyte = ks * 64 + ((lane >> 4) << 4);
      typename MF::frag pf[2], vf[4];
#pragma unroll
      for (int f = 0; f < 2; ++f)
        pf[f] = *(const typename MF::frag*)(Pt +
                 swz(f * 16 + (lane & 15), kbyte));
#pragma unroll
      for (int f = 0; f < 4; ++f)
        vf[f] = *(const typename MF::frag*)(Vk +
                 swz(f * 16 + (lane & 15), kbyte));
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          oacc[i][j] = MF::run(pf[i], vf[j], oacc[i][j]);
    }
  }

  // ---- store out[b*S + row][h*D + d] ----
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int d = j * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = qrow0 + i * 16 + ((lane >> 4) << 2) + r;
        out[((int64_t)b * S + row) * hid + h * D + d] = (T)oacc[i][j][r];
      }
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
                      int H, int D, float scale, hipStream_t stream) {
  if (S != 128 || D != 64)
    throw std::runtime_error("attention: only S=128, D=64 supported (BERT-base seq128)");
  dim3 grid(B * H);
  dim3 block(256);
  if (dtype == 0)
    hipLaunchKernelGGL((attention_kernel<_Float16>), grid, block, 0, stream,
                       (const _Float16*)qkv, (_Float16*)out, B, S, H, D,
                       scale);
  else
    hipLaunchKernelGGL((attention_kernel<__bf16>), grid, block, 0, stream,
                       (const __bf16*)qkv, (__bf16*)out, B, S, H, D, scale);
}

}  // namespace trtlab
