// trtlab_amd — kernel launcher declarations (implemented in kernels/*.hip).
#pragma once
#include <hip/hip_runtime.h>
#include <cstdint>

namespace trtlab {

// dtype: 0 = fp16, 1 = bf16, 2 = int8 (symmetric, per-channel weights)
void launch_gemm_bt(int dtype, const void* A, const void* B, void* C,
                    const float* scale, const float* bias, const void* residual,
                    float res_scale, int M, int N, int K, int64_t lda,
                    int64_t ldb, int64_t ldc, int epi, hipStream_t stream,
                    int tile = 0, void* scratch = nullptr,
                    float out_scale = 1.0f);
size_t gemm_scratch_bytes(int M, int N, int K);

void launch_conv2d(int dtype, const void* in, const void* Wt, void* out,
                   const float* scale, const float* bias, const void* residual,
                   const void* zero_page, int Nb, int H, int W, int C,
                   int Cout, int KH, int KW, int sh, int sw, int ph, int pw,
                   int epi, hipStream_t stream, int tile = 0,
                   void* scratch = nullptr, float res_scale = 1.0f);
size_t conv_scratch_bytes(int Nb, int H, int W, int C, int Cout, int KH,
                          int KW, int sh, int sw, int ph, int pw);

void launch_maxpool2d(int dtype, const void* in, void* out, int Nb, int H,
                      int W, int C, int KH, int KW, int sh, int sw, int ph,
                      int pw, hipStream_t stream);
void launch_avgpool2d(int dtype, const void* in, void* out, int Nb, int H,
                      int W, int C, int KH, int KW, int sh, int sw, int ph,
                      int pw, hipStream_t stream);
void launch_gavgpool(int dtype, const void* in, void* out, int Nb, int HW,
                     int C, hipStream_t stream);

void launch_softmax_rows(int dtype, const void* in, void* out, int M, int N,
                         int64_t ld, hipStream_t stream);
// q_out (optional): fused fp8-e4m3 quantized copy of the output row at
// scale q_scale (producer-side quantization for fp8 projections).
// mx_codes/mx_scales + mx_mode (4 = MXFP4, 8 = MXFP8): producer-fused
// OCP MX quantization of the normalized row (codes + e8m0 block scales),
// layouts identical to quantize_mxfp4/8.
void launch_layernorm(int dtype, const void* in, const float* gamma,
                      const float* beta, void* out, int M, int N, int64_t ld,
                      float eps, hipStream_t stream, void* q_out = nullptr,
                      float q_scale = 0.f, void* mx_codes = nullptr,
                      void* mx_scales = nullptr, int mx_mode = 0);
void launch_add_layernorm(int dtype, const void* x, const void* res,
                          const float* gamma, const float* beta, void* out,
                          void* sum_out, int M, int N, int64_t ld, float eps,
                          hipStream_t stream, void* q_out = nullptr,
                          float q_scale = 0.f, void* mx_codes = nullptr,
                          void* mx_scales = nullptr, int mx_mode = 0);

void launch_rmsnorm(int dtype, const void* in, const float* gamma, void* out,
                    int M, int N, int64_t ld, float eps, hipStream_t stream);
void launch_add_rmsnorm(int dtype, const void* x, const void* res,
                        const float* gamma, void* out, void* sum_out, int M,
                        int N, int64_t ld, float eps, hipStream_t stream);
void launch_silu_mul(int dtype, const void* a, const void* b, void* out,
                     int64_t n, hipStream_t stream);
// RoPE on the q/k blocks of fused qkv rows. pos_dev = null: pos = row % S
// (full-sequence forward). pos_dev set, chunk = 0: pos = pos_dev[row]
// (decode step, M = B slots). chunk = K > 0: pos = pos_dev[row / K] +
// row % K (speculative verify chunk, rows grouped K-per-slot).
void launch_rope(int dtype, void* qkv, const void* pos_dev, int M, int S,
                 int H, int D, float theta, hipStream_t stream,
                 int chunk = 0);
// Fused ResNet bottleneck tail: conv3x3(s1,p1,Cm)+BN+ReLU feeding
// conv1x1(Cm->Co)+BN+residual+ReLU in ONE kernel (intermediate stays in
// LDS). Cm in {64,128}; fp16 only.
void launch_bottleneck_tail(int dtype, const void* in, const void* W1,
                            const void* W2, void* out, const float* s1,
                            const float* b1, const float* s2, const float* b2,
                            const void* residual, const void* zero_page,
                            int Nb, int H, int W, int Cm, int Co,
                            hipStream_t stream);
void launch_elementwise(int dtype, int op, const void* a, const void* b,
                        void* out, int64_t n, hipStream_t stream);
// Per-channel affine (+ReLU): standalone batchnorm (DenseNet pre-act).
void launch_channel_affine(int dtype, const void* x, void* out,
                           const float* s, const float* b, int64_t M, int C,
                           bool relu, hipStream_t stream);
// Row-wise argmax (greedy decode head): fp16 [M, V] -> int32 [M].
void launch_argmax_rows(const void* x, void* out, int M, int V,
                        hipStream_t stream);
// Gumbel-max categorical sampling from softmax(logits/temps[row]);
// temps[row] <= 0 -> plain argmax. Noise keyed (seeds[row], pos[row], c).
void launch_gumbel_argmax_rows(const void* x, void* out, const void* temps,
                               const void* seeds, const void* pos, int M,
                               int V, hipStream_t stream);
void launch_clip(int dtype, const void* in, void* out, int64_t n, float mn,
                 float mx, hipStream_t stream);
void launch_transpose2d(int dtype, const void* in, void* out, int M, int N,
                        hipStream_t stream);
void launch_copy2d(int dtype, const void* src, void* dst, int64_t M, int C,
                   int ldd, int coff, hipStream_t stream);
void launch_channel_pad(int dtype, const void* in, void* out, int64_t M,
                        int Cin, int Cpad, hipStream_t stream);
void launch_cast(int dtype, bool to_f32, const void* in, void* out, int64_t n,
                 hipStream_t stream);

// out_dtype 3 = fp8-e4m3 output at out_scale (fused quantization for the
// following projection); otherwise the compute dtype.
void launch_attention(int dtype, const void* qkv, void* out, int B, int S,
                      int H, int D, float scale, hipStream_t stream,
                      int out_dtype = -1, float out_scale = 1.0f,
                      const void* seqlens = nullptr, int causal = 0);

// lens[b] = count of non-pad tokens in right-padded ids (>=1); feeds the
// variable-length attention mask.
void launch_seqlens(const void* ids, void* lens, int B, int S, int pad_id,
                    hipStream_t stream);

// MXFP8 (OCP MX: fp8-e4m3 elements + e8m0 per-32-block scales) GEMM on
// the CDNA4 scaled MFMA 16x16x128. B transposed [N][K]; scales u8
// [M][K/32] / [N][K/32]; C fp32.
void launch_mx_frag_dump(const void* A, void* out, int K, hipStream_t s);
void launch_quantize_mxfp4(const void* x, void* codes, void* scales,
                           int64_t m, int64_t k, hipStream_t stream);
void launch_gemm_mxfp4(const void* A, const void* B, const void* Sa,
                       const void* Sb, void* C, int M, int N, int K,
                       hipStream_t stream, int out_dtype = 2,
                       int epi = 0, const float* scale = nullptr,
                       const float* bias = nullptr);
void launch_mx4_probe(const void* A, const void* B, const void* Sa,
                      const void* Sb, void* D, hipStream_t stream);
void launch_mx_probe(const void* A, const void* B, const void* Sa,
                     const void* Sb, void* D, hipStream_t stream);
void launch_gemm_mxfp8(const void* A, const void* B, const void* Sa,
                       const void* Sb, void* C, int M, int N, int K,
                       hipStream_t stream, int out_dtype = 2, int epi = 0,
                       const float* scale = nullptr,
                       const float* bias = nullptr);
void launch_quantize_mxfp8(const void* x, void* codes, void* scales,
                           int64_t m, int64_t k, hipStream_t stream);

// ---- incremental decode (KV cache; see csrc/kernels/decode.hip) ----
void launch_kv_append(const void* qkv, void* kcache, void* vcache,
                      const void* pos, int B, int H, int smax,
                      hipStream_t stream, int D = 64);
void launch_kv_append_range(const void* qkv, void* kcache, void* vcache,
                            int B, int H, int P, int smax,
                            hipStream_t stream, int D = 64);
void launch_decode_attention(const void* qkv, const void* kcache,
                             const void* vcache, void* out, const void* pos,
                             int B, int H, int smax, float scale,
                             hipStream_t stream, int D = 64);
void launch_decode_embed(const void* ids, const void* tok, const void* posemb,
                         void* out, const void* pos, int B, int hidden,
                         hipStream_t stream);
// speculative-decoding verification chunk (see decode.hip)
void launch_kv_append_chunk(const void* qkv, void* kcache, void* vcache,
                            const void* pos, int B, int H, int K, int smax,
                            hipStream_t stream, int D = 64);
void launch_chunk_attention(const void* qkv, const void* kcache,
                            const void* vcache, void* out, const void* pos,
                            int B, int H, int K, int smax, float scale,
                            hipStream_t stream, int D = 64);
void launch_chunk_embed(const void* ids, const void* tok, const void* posemb,
                        void* out, const void* pos, int B, int K, int smax,
                        int hidden, hipStream_t stream);
void launch_advance_pos(void* pos, int B, int smax, hipStream_t stream);
// paged KV cache (vLLM-style block tables; see decode.hip)
void launch_kv_append_chunk_paged(const void* qkv, void* kpool, void* vpool,
                                  const void* table, const void* pos, int B,
                                  int H, int K, int max_pages,
                                  hipStream_t stream, int D = 64);
void launch_kv_append_range_paged(const void* qkv, void* kpool, void* vpool,
                                  const void* table, int B, int H, int P,
                                  int max_pages, hipStream_t stream,
                                  int D = 64);
void launch_chunk_attention_paged(const void* qkv, const void* kpool,
                                  const void* vpool, void* out,
                                  const void* table, const void* pos, int B,
                                  int H, int K, int max_pages, float scale,
                                  hipStream_t stream, int D = 64);
void launch_kv_append_paged(const void* qkv, void* kpool, void* vpool,
                            const void* table, const void* pos, int B, int H,
                            int max_pages, hipStream_t stream, int D = 64);
void launch_decode_attention_paged(const void* qkv, const void* kpool,
                                   const void* vpool, void* out,
                                   const void* table, const void* pos, int B,
                                   int H, int max_pages, float scale,
                                   hipStream_t stream, int D = 64);
// fused decode GEMM: prologue 1=LN 2=ADD_LN(+h_out) 3=EMBED_LN; epilogue
// 0=bias 1=bias+gelu 2=bias+KV-scatter (see decode.hip)
void launch_decode_gemm_fused(int pro, int epi, const void* x, const void* r,
                              void* h_out, const float* gamma,
                              const float* beta, const void* Bw,
                              const float* bias, void* C, const void* ids,
                              const void* tok, const void* posemb,
                              const void* pos, void* kcache, void* vcache,
                              int M, int N, int K, int heads, int smax,
                              float eps, hipStream_t stream);

void launch_embedding(int dtype, const void* ids, const void* tok,
                      const void* pos, const void* seg, const void* segids,
                      void* out, int M, int S, int H, hipStream_t stream);

// int8 quantization staging: out_i8 = clamp(round(in_f16 / scale)),
// and the inverse. n % 8 == 0.
// fmt: 0 = int8 (round-to-int codes), 1 = fp8 e4m3 (continuous, sat 448)
void launch_quantize(const void* in_f16, void* out_q, int64_t n, float scale,
                     hipStream_t stream, int fmt = 0);
void launch_dequant(const void* in_q, void* out_f16, int64_t n, float scale,
                    hipStream_t stream, int fmt = 0);

}  // namespace trtlab
