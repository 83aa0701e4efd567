// trtlab_amd — BERT embedding gather (gfx950): out[m] = tok[ids[m]] +
// pos[m % S] (+ seg[segids[m]]). SURVEY.md §2.8 item 8 (embedding gather).
// Gather rows are contiguous fp16 — vectorized 16-B copies; uncoalesced by
// nature across rows, L2/L3 absorbs the table re-reads (vocab tables are
// << 256 MiB).
#include "../common.h"

namespace trtlab {

template <typename T>
__global__ void embedding_kernel(const int* __restrict__ ids,
                                 const T* __restrict__ tok,
                                 const T* __restrict__ pos,
                                 const T* __restrict__ seg,
                                 const int* __restrict__ segids,
                                 T* __restrict__ out, int M, int S, int H) {
  int lane = threadIdx.x & 63;
  int row = (blockIdx.x * (blockDim.x >> 6)) + (threadIdx.x >> 6);
  if (row >= M) return;
  const T* trow = tok + (int64_t)ids[row] * H;
  const T* prow = pos + (int64_t)(row % S) * H;
  const T* srow = seg ? seg + (int64_t)(segids ? segids[row] : 0) * H : nullptr;
  T* orow = out + (int64_t)row * H;
  for (int c = lane * 8; c < H; c += 64 * 8) {
    short4v t0 = *(const short4v*)(trow + c);
    short4v t1 = *(const short4v*)(trow + c + 4);
    short4v p0 = *(const short4v*)(prow + c);
    short4v p1 = *(const short4v*)(prow + c + 4);
    T r[8];
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      r[j] = (T)((float)((const T*)&t0)[j] + (float)((const T*)&p0)[j]);
      r[4 + j] = (T)((float)((const T*)&t1)[j] + (float)((const T*)&p1)[j]);
    }
    if (srow) {
#pragma unroll
      for (int j = 0; j < 8; ++j) r[j] = (T)((float)r[j] + (float)srow[c + j]);
    }
    *(short4v*)(orow + c) = *(const short4v*)&r[0];
    *(short4v*)(orow + c + 4) = *(const short4v*)&r[4];
  }
}

void launch_embedding(int dtype, const void* ids, const void* tok,
                      const void* pos, const void* seg, const void* segids,
                      void* out, int M, int S, int H, hipStream_t stream) {
  if (H % 8 != 0) throw std::runtime_error("embedding: H % 8 != 0");
  dim3 grid((unsigned)cdiv(M, 4));
  dim3 block(256);
  if (dtype == 0)
    hipLaunchKernelGGL((embedding_kernel<_Float16>), grid, block, 0, stream,
                       (const int*)ids, (const _Float16*)tok,
                       (const _Float16*)pos, (const _Float16*)seg,
                       (const int*)segids, (_Float16*)out, M, S, H);
  else
    hipLaunchKernelGGL((embedding_kernel<__bf16>), grid, block, 0, stream,
                       (const int*)ids, (const __bf16*)tok,
                       (const __bf16*)pos, (const __bf16*)seg,
                       (const int*)segids, (__bf16*)out, M, S, H);
}

}  // namespace trtlab
