// trtlab_amd — native runtime: memory primitives + graph-captured executor.
#pragma once
#include "../common.h"
#include "hybrid.h"

#include <array>
#include <atomic>
#include <map>
#include <memory>
#include <mutex>
#include <set>
#include <vector>

namespace trtlab {

// ------------------------------------------------------------------ memory
void* device_malloc(size_t bytes, int device);
void device_free(void* p, size_t bytes);
void* pinned_malloc(size_t bytes);
void pinned_free(void* p, size_t bytes);
int64_t device_bytes_in_use();
int64_t pinned_bytes_in_use();
void* huge_malloc(size_t bytes, bool pin, bool* hugetlb_out);
void huge_free(void* p, size_t bytes, bool pin);
int64_t huge_bytes_in_use();

class BlockPool {
 public:
  BlockPool(size_t block_bytes, int count, int device);
  ~BlockPool();
  void* acquire();
  void release(void* p);
  int available();
  size_t block_bytes() const { return block_bytes_; }
  int total() const { return total_; }

 private:
  size_t block_bytes_;
  int device_;
  char* base_;
  int total_;
  std::vector<char*> free_;
  // hybrid spin-then-futex lock (reference hybrid_mutex role): the pool's
  // acquire/release critical sections are a few instructions — spinning
  // keeps the hot dispatch path out of the kernel
  HybridMutex mu_;
};

// Growing best-fit device allocator for dynamic workloads (multi-model
// arena sharing, decode KV growth) — reference bfit_allocator.h:121 +
// block_arena.h:177. All bookkeeping host-side (device memory cannot hold
// list nodes: the reference's block_list_oob insight).
class DeviceArena {
 public:
  struct Stats {
    size_t capacity = 0, in_use = 0, high_water = 0, largest_free = 0;
    size_t free_nodes = 0, live_allocs = 0;
    uint64_t histogram[48] = {0};  // log2 allocation-size buckets
  };

  DeviceArena(int device, size_t initial_bytes, size_t max_bytes = 0,
              size_t growth_bytes = 0);
  ~DeviceArena();
  DeviceArena(const DeviceArena&) = delete;
  DeviceArena& operator=(const DeviceArena&) = delete;

  void* allocate(size_t bytes, size_t align = 256);
  void deallocate(void* p);
  Stats stats() const;
  int device() const { return device_; }

 private:
  struct Slab {
    char* base;
    size_t bytes;
  };
  static inline size_t round_up(int64_t a, int64_t b) {
    return (size_t)(((a + b - 1) / b) * b);
  }
  void grow(size_t need);
  void insert_free(char* p, size_t bytes);
  void erase_size(char* p, size_t bytes);

  int device_;
  size_t max_bytes_, growth_bytes_;
  size_t capacity_ = 0, in_use_ = 0, high_water_ = 0;
  std::vector<Slab> slabs_;
  std::map<char*, size_t> by_addr_;              // free nodes by address
  std::set<std::pair<size_t, char*>> by_size_;   // free nodes by size
  std::map<char*, size_t> live_;                 // outstanding allocations
  uint64_t hist_[48] = {0};
  mutable std::mutex mu_;
};

// ---------------------------------------------------------------- executor
// Op kinds executed by the engine (planned host-side in Python).
enum OpKind : int {
  kConv2d = 0,
  kGemmBt = 1,
  kMaxPool = 2,
  kGAvgPool = 3,
  kSoftmax = 4,
  kLayerNorm = 5,
  kAddLayerNorm = 6,
  kElementwise = 7,
  kChannelPad = 8,
  kAttention = 9,
  kQuantize = 10,   // fp16 -> int8 (symmetric, scale)
  kDequant = 11,    // int8 -> fp16 (scale)
  kEmbedding = 12,  // out = tok[ids] + pos[m%S] (+ seg)
  kAvgPool = 13,
  kSeqLens = 14,  // ids -> per-sequence valid length (varlen attention)
  kQuantMx4 = 15,  // fp16 rows -> MXFP4 codes + e8m0 block scales
  kGemmMx4 = 16,   // MXFP4 x MXFP4 scaled-MFMA GEMM, fp16 out + epilogue
  kQuantMx8 = 17,  // fp16 rows -> MXFP8 (e4m3) codes + e8m0 block scales
  kGemmMx8 = 18,   // MXFP8 x MXFP8 scaled-MFMA GEMM, fp16 out + epilogue
  kClip = 19,        // out = min(max(x, res_scale), q_scale) — ONNX Clip
  kTranspose2D = 20, // out[N][M] = in[M][N]^T (tiled LDS transpose)
  kCopy2D = 21,      // dst[m][epi + c] = src[m][c], dst row stride Cout
  kRMSNorm = 22,     // LLaMA norm: x / rms(x) * gamma (no mean, no beta)
  kSiluMul = 23,     // SwiGLU gate: silu(a) * b
  kRope = 24,        // rotary embedding in-place on qkv q/k blocks
  kBtail = 25,       // fused bottleneck tail: 3x3+BN+ReLU -> 1x1+BN+res+ReLU
  kConst = 26,       // weight-blob constant -> arena tensor (D2D copy)
  kView = 27,        // zero-copy reshape (arena alias; no kernel)
  kChAffine = 28,    // out[m,c] = x[m,c]*s[c]+b[c] (+ReLU via epi=1)
};

struct OpDesc {
  int kind = 0;
  int dtype = 0;  // 0 fp16, 1 bf16
  int epi = 0;    // epilogue enum / elementwise op code
  // arena offsets in bytes (-1 = absent); out3 = MX block-scales output
  int64_t in_off = -1, in2_off = -1, out_off = -1, out2_off = -1;
  int64_t out3_off = -1;
  // weight-blob offsets in bytes (-1 = absent)
  int64_t w_off = -1, scale_off = -1, bias_off = -1;
  int64_t w2_off = -1;  // second weight slab (MX weight scales)
  // gemm
  int M = 0, N = 0, K = 0;
  // conv / pool
  int Nb = 0, H = 0, W = 0, C = 0, Cout = 0, KH = 0, KW = 0, sh = 1, sw = 1,
      ph = 0, pw = 0, HW = 0;
  // norms / elementwise
  float eps = 1e-5f;
  int64_t n_elems = 0;
  // attention
  int B = 0, S = 0, NH = 0, HD = 0;
  int causal = 0;  // decoder-style key > query masking
  float att_scale = 1.0f;
  // int8: residual dequant ratio (s_res/s_out); quantize/dequant scale
  float res_scale = 1.0f;
  float q_scale = 1.0f;
  // autotuned tile override (0 = heuristic; 1..4 = fixed BMxBN config)
  int tile = 0;
  // fork/join dual-stream schedule (ResNet downsample pattern): fork=1
  // launches this op on the context's side stream (its inputs are ready
  // before the main chain it overlaps); join=1 makes the op wait for the
  // pending forked op's event first. Pairs are strictly sequential.
  int fork = 0;
  int join = 0;
};

// One engine I/O binding: a named arena region mirrored by pinned host
// staging in each ExecutionContext. The reference carves N host+device
// addresses per model the same way (trtlab/tensorrt/bindings.h:60-120).
struct BindingDesc {
  int64_t off = -1;   // arena offset in bytes
  size_t bytes = 0;   // binding size
};

// A compiled model: weight blob on device + op list + arena layout.
// Replaces the reference's Model/ICudaEngine (trtlab/tensorrt/model.h:17):
// here the "engine" is an explicit op plan over hand-written CDNA4 kernels.
class Engine {
 public:
  // managed_weights: place the weight blob in hipMallocManaged memory with
  // hipMemAdvise(ReadMostly) + device prefetch — the reference's
  // ManagedRuntime / NvAllocator::use_weights_allocator path
  // (trtlab/tensorrt/src/allocator.cc:12-56). Default off: explicit device
  // memory is faster and the MI355X has 288 GB of HBM3E per GPU.
  Engine(int device, const void* weights, size_t weight_bytes,
         size_t arena_bytes, std::vector<OpDesc> ops,
         std::vector<BindingDesc> inputs, std::vector<BindingDesc> outputs,
         bool managed_weights = false);
  ~Engine();

  int device() const { return device_; }
  size_t arena_bytes() const { return arena_bytes_; }
  size_t scratch_bytes() const { return scratch_bytes_; }
  const std::vector<BindingDesc>& inputs() const { return inputs_; }
  const std::vector<BindingDesc>& outputs() const { return outputs_; }
  // Primary (first-input / first-output) conveniences.
  size_t input_bytes() const { return inputs_[0].bytes; }
  size_t output_bytes() const { return outputs_[0].bytes; }
  int64_t input_off() const { return inputs_[0].off; }
  int64_t output_off() const { return outputs_[0].off; }
  const std::vector<OpDesc>& ops() const { return ops_; }
  const char* weights() const { return (const char*)weights_; }
  const char* zero_page() const { return (const char*)zero_page_; }
  // Overwrite the device weight blob (RCCL broadcast target, tests).
  void upload_weights(const void* src, size_t bytes);
  uintptr_t weights_ptr() const { return (uintptr_t)weights_; }
  size_t weight_bytes() const { return weight_bytes_; }

 private:
  int device_;
  void* weights_ = nullptr;
  size_t weight_bytes_ = 0;
  bool managed_weights_ = false;
  void* zero_page_ = nullptr;
  size_t arena_bytes_;
  size_t scratch_bytes_ = 0;  // split-K slab workspace (max over ops)
  std::vector<OpDesc> ops_;
  std::vector<BindingDesc> inputs_, outputs_;
};

// Per-request execution context: private stream + activation arena + pinned
// staging + captured hipGraph. Mirrors the reference's
// StaticSingleModelGraphWorkspace (trtlab/tensorrt/src/workspace.cc:21-75):
// allocate bindings/scratch, warm up, capture enqueue into a graph, replay.
class ExecutionContext {
 public:
  // external_arena: optional caller-owned device pointer for the
  // activation arena (e.g. carved from a shared DeviceArena so multiple
  // models serve from one pool); 0 = allocate privately.
  ExecutionContext(std::shared_ptr<Engine> engine,
                   uintptr_t external_arena = 0);
  ~ExecutionContext();

  // Raw host staging buffers (pinned), one per binding. Index 0 keeps the
  // single-binding API shape.
  uintptr_t host_input_ptr(int i = 0) const {
    return (uintptr_t)(h_in_ + in_hoff_[i]);
  }
  uintptr_t host_output_ptr(int i = 0) const {
    return (uintptr_t)(h_out_ + out_hoff_[i]);
  }
  uintptr_t arena_ptr() const { return (uintptr_t)arena_; }

  void enqueue_all(hipStream_t s);  // H2D + ops + D2H on stream s
  void capture();                   // warm-up + hipGraph capture
  void launch();                    // graph launch (or eager enqueue)
  void synchronize();
  bool ready();  // hipStreamQuery == success

  // Per-stage timing (reference TimedBenchmarkWorkspace,
  // workspace.cc:128-168: 4 events bracket H2D / compute / D2H).
  // Events are recorded inside the captured graph; call after synchronize.
  void set_timing(bool enable) { timing_ = enable; }
  std::array<float, 3> stage_times_ms() const;

 private:
  std::shared_ptr<Engine> eng_;
  char* arena_ = nullptr;
  bool owns_arena_ = true;
  char* scratch_ = nullptr;
  // One pinned slab per direction, carved per binding (in_hoff_[i] = host
  // offset of binding i) — the reference's Buffers/Bindings carving
  // (trtlab/tensorrt/src/buffers.cc, bindings.h:60-120).
  char* h_in_ = nullptr;
  char* h_out_ = nullptr;
  std::vector<size_t> in_hoff_, out_hoff_;
  size_t h_in_bytes_ = 0, h_out_bytes_ = 0;
  hipStream_t stream_{};
  // fork/join side stream (captured as a forked branch of the main
  // stream's graph): downsample convs overlap the main bottleneck chain
  hipStream_t side_{};
  char* scratch2_ = nullptr;  // private split-K slab for forked ops
  std::vector<hipEvent_t> fork_ev_;  // 2 events per fork op (fork + join)
  hipGraph_t graph_{};
  hipGraphExec_t graph_exec_{};
  bool captured_ = false;
  bool timing_ = false;
  hipEvent_t ev_[4] = {nullptr, nullptr, nullptr, nullptr};
};

}  // namespace trtlab
