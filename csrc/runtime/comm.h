// trtlab_amd — owned RCCL collective layer over xGMI.
//
// The reference has NO collective backend (MPI appears only
// as barriers for MPS benchmarks, examples/00_TensorRT/inference.cc:37-42).
// This is the MI355X-native addition: a first-class communicator that owns
// ncclComm lifecycle, runs collectives on caller-supplied HIP streams (so
// they order against the engine's own streams), and fails loudly on RCCL
// errors. One communicator per process, one process per GPU; xGMI is
// point-to-point (7 links x ~153 GB/s per GPU) so fused whole-blob
// transfers beat many small messages.
#pragma once
#include "../common.h"

#include <string>

// Forward-declare the RCCL handle so rccl.h stays out of every TU.
typedef struct ncclComm* ncclComm_t;

namespace trtlab {

// Reduction ops (subset we use; keep in sync with ncclRedOp_t).
enum CommRedOp : int { kCommSum = 0, kCommProd = 1, kCommMax = 2,
                       kCommMin = 3, kCommAvg = 4 };

// Element types for typed collectives.
enum CommDType : int { kCommU8 = 0, kCommF16 = 1, kCommBF16 = 2,
                       kCommF32 = 3, kCommF64 = 4, kCommI32 = 5 };

class Communicator {
 public:
  // 128-byte ncclUniqueId produced by rank 0 and shared out-of-band
  // (file/TCP rendezvous lives in Python: trtlab_amd.parallel).
  static std::string unique_id();

  // Collective constructor: every rank of the clique must call this with
  // the same uid. Blocks until the clique is connected. device = the HIP
  // device this rank drives (ranks may share a device: RCCL supports
  // multi-rank-per-GPU, which is how the 2-rank proof runs on a 1-GPU box).
  Communicator(int rank, int world, const std::string& uid, int device);
  ~Communicator();

  Communicator(const Communicator&) = delete;
  Communicator& operator=(const Communicator&) = delete;

  int rank() const { return rank_; }
  int world() const { return world_; }
  int device() const { return device_; }

  // Stream-ordered collectives on DEVICE pointers. stream 0 = the
  // communicator's own stream.
  void broadcast(void* ptr, size_t bytes, int root, hipStream_t s);
  void all_reduce(void* ptr, size_t count, int dtype, int op, hipStream_t s);
  void all_gather(const void* send, void* recv, size_t bytes_per_rank,
                  hipStream_t s);
  void reduce_scatter(const void* send, void* recv, size_t count_per_rank,
                      int dtype, int op, hipStream_t s);
  void send(const void* ptr, size_t bytes, int peer, hipStream_t s);
  void recv(void* ptr, size_t bytes, int peer, hipStream_t s);

  // Host conveniences (stage through the comm's device scratch + sync):
  // the bench's barrier + MAX-over-ranks timing choreography.
  void barrier();
  double all_reduce_scalar(double v, int op);

  void stream_synchronize();

 private:
  hipStream_t resolve(hipStream_t s) const {
    return s ? s : stream_;
  }
  ncclComm_t comm_ = nullptr;
  int rank_, world_, device_;
  hipStream_t stream_{};  // comm-owned side stream
  void* scratch_ = nullptr;  // 256-B device scratch for scalar staging
};

}  // namespace trtlab
