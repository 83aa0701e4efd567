// trtlab_amd — fused multi-head self-attention (BERT encoder path).
// Implemented after the ResNet path is proven on hardware; the launcher
// exists so the executor op table is complete. SURVEY.md §2.8 item 3.
#include "../common.h"

namespace trtlab {

void launch_attention(int dtype, const void* qkv, void* out, int B, int S,
                      int H, int D, float scale, hipStream_t stream) {
  (void)dtype; (void)qkv; (void)out; (void)B; (void)S; (void)H; (void)D;
  (void)scale; (void)stream;
  throw std::runtime_error("attention kernel: not implemented yet");
}

}  // namespace trtlab
