// trtlab_amd — pooling kernels (NHWC fp16/bf16), gfx950.
// Memory-bound: vectorized 8-wide half loads (G13), grid-stride loops.
// Covers the reference's TensorRT-internal pool ops.
#include <hip/hip_fp8.h>

#include "../common.h"

namespace trtlab {

template <typename T>
__global__ void maxpool2d_kernel(const T* __restrict__ in, T* __restrict__ out,
                                 int Nb, int H, int W, int C, int KH, int KW,
                                 int sh, int sw, int ph, int pw, int OH,
                                 int OW) {
  // one thread = one (n, oh, ow, 8-channel group)
  int cg = C >> 3;
  int64_t total = (int64_t)Nb * OH * OW * cg;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    int c8 = (int)(idx % cg);
    int64_t q = idx / cg;
    int ow = (int)(q % OW);
    q /= OW;
    int oh = (int)(q % OH);
    int n = (int)(q / OH);
    float best[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) best[j] = -3.0e38f;
    for (int kh = 0; kh < KH; ++kh) {
      int ih = oh * sh - ph + kh;
      if ((unsigned)ih >= (unsigned)H) continue;
      for (int kw = 0; kw < KW; ++kw) {
        int iw = ow * sw - pw + kw;
        if ((unsigned)iw >= (unsigned)W) continue;
        const T* p = in + ((((int64_t)n * H + ih) * W + iw) * C) + c8 * 8;
        T tmp[8];
        if constexpr (sizeof(T) == 2) {
          *(short4v*)&tmp[0] = *(const short4v*)p;
          *(short4v*)&tmp[4] = *(const short4v*)(p + 4);
        } else {
          *(uint2*)&tmp[0] = *(const uint2*)p;
        }
#pragma unroll
        for (int j = 0; j < 8; ++j)
          best[j] = fmaxf(best[j], (float)tmp[j]);
      }
    }
    T* o = out + ((((int64_t)n * OH + oh) * OW + ow) * C) + c8 * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (T)best[j];  // exact for int8 maxima
  }
}

// Windowed average pool (NHWC, count_include_pad=False — torch default).
template <typename T>
__global__ void avgpool2d_kernel(const T* __restrict__ in, T* __restrict__ out,
                                 int Nb, int H, int W, int C, int KH, int KW,
                                 int sh, int sw, int ph, int pw, int OH,
                                 int OW) {
  int cg = C >> 3;
  int64_t total = (int64_t)Nb * OH * OW * cg;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
       idx < total; idx += (int64_t)gridDim.x * blockDim.x) {
    int c8 = (int)(idx % cg);
    int64_t q = idx / cg;
    int ow = (int)(q % OW);
    q /= OW;
    int oh = (int)(q % OH);
    int n = (int)(q / OH);
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    int cnt = 0;
    for (int kh = 0; kh < KH; ++kh) {
      int ih = oh * sh - ph + kh;
      if ((unsigned)ih >= (unsigned)H) continue;
      for (int kw = 0; kw < KW; ++kw) {
        int iw = ow * sw - pw + kw;
        if ((unsigned)iw >= (unsigned)W) continue;
        const T* p = in + ((((int64_t)n * H + ih) * W + iw) * C) + c8 * 8;
        T tmp[8];
        if constexpr (sizeof(T) == 2) {
          *(short4v*)&tmp[0] = *(const short4v*)p;
          *(short4v*)&tmp[4] = *(const short4v*)(p + 4);
        } else {
          *(uint2*)&tmp[0] = *(const uint2*)p;
        }
#pragma unroll
        for (int j = 0; j < 8; ++j) acc[j] += (float)tmp[j];
        ++cnt;
      }
    }
    float inv = cnt ? 1.0f / cnt : 0.0f;
    T* o = out + ((((int64_t)n * OH + oh) * OW + ow) * C) + c8 * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (T)(acc[j] * inv);
  }
}

void launch_avgpool2d(int dtype, const void* in, void* out, int Nb, int H,
                      int W, int C, int KH, int KW, int sh, int sw, int ph,
                      int pw, hipStream_t stream) {
  if (C % 8 != 0) throw std::runtime_error("avgpool2d: C % 8 != 0");
  int OH = (H + 2 * ph - KH) / sh + 1;
  int OW = (W + 2 * pw - KW) / sw + 1;
  int64_t total = (int64_t)Nb * OH * OW * (C / 8);
  int blocks = (int)std::min<int64_t>(cdiv(total, 256), 2048);
  if (dtype == 0)
    hipLaunchKernelGGL((avgpool2d_kernel<_Float16>), dim3(blocks), dim3(256),
                       0, stream, (const _Float16*)in, (_Float16*)out, Nb, H,
                       W, C, KH, KW, sh, sw, ph, pw, OH, OW);
  else if (dtype == 1)
    hipLaunchKernelGGL((avgpool2d_kernel<__bf16>), dim3(blocks), dim3(256), 0,
                       stream, (const __bf16*)in, (__bf16*)out, Nb, H, W, C,
                       KH, KW, sh, sw, ph, pw, OH, OW);
  else
    throw std::runtime_error("avgpool2d: fp16/bf16 only");
}

// Global average pool: in [Nb, HW, C] -> out [Nb, C], fp32 accumulate.
template <typename T>
__global__ void gavgpool_kernel(const T* __restrict__ in, T* __restrict__ out,
                                int Nb, int HW, int C) {
  int cg = C >> 3;
  int total = Nb * cg;
  for (int idx = blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += gridDim.x * blockDim.x) {
    int c8 = idx % cg;
    int n = idx / cg;
    float acc[8] = {0, 0, 0, 0, 0, 0, 0, 0};
    const T* base = in + ((int64_t)n * HW * C) + c8 * 8;
    for (int i = 0; i < HW; ++i) {
      short4v v0 = *(const short4v*)(base + (int64_t)i * C);
      short4v v1 = *(const short4v*)(base + (int64_t)i * C + 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        acc[j] += (float)((const T*)&v0)[j];
        acc[4 + j] += (float)((const T*)&v1)[j];
      }
    }
    float inv = 1.0f / (float)HW;
    T* o = out + (int64_t)n * C + c8 * 8;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = (T)(acc[j] * inv);
  }
}

template <typename T>
static void launch_maxpool2d_t(const void* in, void* out, int Nb, int H, int W,
                               int C, int KH, int KW, int sh, int sw, int ph,
                               int pw, hipStream_t stream) {
  int OH = (H + 2 * ph - KH) / sh + 1;
  int OW = (W + 2 * pw - KW) / sw + 1;
  int64_t total = (int64_t)Nb * OH * OW * (C / 8);
  int blocks = (int)std::min<int64_t>(cdiv(total, 256), 2048);
  hipLaunchKernelGGL((maxpool2d_kernel<T>), dim3(blocks), dim3(256), 0, stream,
                     (const T*)in, (T*)out, Nb, H, W, C, KH, KW, sh, sw, ph,
                     pw, OH, OW);
}

void launch_maxpool2d(int dtype, const void* in, void* out, int Nb, int H,
                      int W, int C, int KH, int KW, int sh, int sw, int ph,
                      int pw, hipStream_t stream) {
  if (C % 8 != 0) throw std::runtime_error("maxpool2d: C % 8 != 0");
  if (dtype == 0)
    launch_maxpool2d_t<_Float16>(in, out, Nb, H, W, C, KH, KW, sh, sw, ph, pw, stream);
  else if (dtype == 1)
    launch_maxpool2d_t<__bf16>(in, out, Nb, H, W, C, KH, KW, sh, sw, ph, pw, stream);
  else if (dtype == 2)
    launch_maxpool2d_t<int8_t>(in, out, Nb, H, W, C, KH, KW, sh, sw, ph, pw, stream);
  else
    launch_maxpool2d_t<__hip_fp8_e4m3>(in, out, Nb, H, W, C, KH, KW, sh, sw,
                                       ph, pw, stream);
}

void launch_gavgpool(int dtype, const void* in, void* out, int Nb, int HW,
                     int C, hipStream_t stream) {
  if (C % 8 != 0) throw std::runtime_error("gavgpool: C % 8 != 0");
  int total = Nb * (C / 8);
  int blocks = std::min((int)cdiv(total, 256), 2048);
  if (blocks < 1) blocks = 1;
  if (dtype == 0)
    hipLaunchKernelGGL((gavgpool_kernel<_Float16>), dim3(blocks), dim3(256), 0,
                       stream, (const _Float16*)in, (_Float16*)out, Nb, HW, C);
  else
    hipLaunchKernelGGL((gavgpool_kernel<__bf16>), dim3(blocks), dim3(256), 0,
                       stream, (const __bf16*)in, (__bf16*)out, Nb, HW, C);
}

}  // namespace trtlab
