// trtlab_amd — owned RCCL collective layer (see comm.h).
#include "comm.h"

#include <rccl/rccl.h>

#include <cstring>

#define TRT_NCCL_CHECK(expr)                                                  \
  do {                                                                        \
    ncclResult_t _r = (expr);                                                 \
    if (_r != ncclSuccess) {                                                  \
      throw std::runtime_error(std::string("RCCL error: ") +                  \
                               ncclGetErrorString(_r) + " at " + __FILE__ +   \
                               ":" + std::to_string(__LINE__) + " in " +      \
                               #expr);                                        \
    }                                                                         \
  } while (0)

namespace trtlab {

namespace {

ncclRedOp_t red_op(int op) {
  switch (op) {
    case kCommSum: return ncclSum;
    case kCommProd: return ncclProd;
    case kCommMax: return ncclMax;
    case kCommMin: return ncclMin;
    case kCommAvg: return ncclAvg;
    default: throw std::runtime_error("bad reduction op");
  }
}

ncclDataType_t nccl_dtype(int dt) {
  switch (dt) {
    case kCommU8: return ncclUint8;
    case kCommF16: return ncclFloat16;
    case kCommBF16: return ncclBfloat16;
    case kCommF32: return ncclFloat32;
    case kCommF64: return ncclFloat64;
    case kCommI32: return ncclInt32;
    default: throw std::runtime_error("bad comm dtype");
  }
}

size_t dtype_size(int dt) {
  switch (dt) {
    case kCommU8: return 1;
    case kCommF16: case kCommBF16: return 2;
    case kCommF32: case kCommI32: return 4;
    case kCommF64: return 8;
    default: throw std::runtime_error("bad comm dtype");
  }
}

}  // namespace

std::string Communicator::unique_id() {
  ncclUniqueId id;
  TRT_NCCL_CHECK(ncclGetUniqueId(&id));
  return std::string(id.internal, NCCL_UNIQUE_ID_BYTES);
}

Communicator::Communicator(int rank, int world, const std::string& uid,
                           int device)
    : rank_(rank), world_(world), device_(device) {
  if ((int)uid.size() != NCCL_UNIQUE_ID_BYTES)
    throw std::runtime_error("Communicator: unique id must be 128 bytes, got " +
                             std::to_string(uid.size()));
  TRT_HIP_CHECK(hipSetDevice(device_));
  TRT_HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
  TRT_HIP_CHECK(hipMalloc(&scratch_, 256));
  ncclUniqueId id;
  std::memcpy(id.internal, uid.data(), NCCL_UNIQUE_ID_BYTES);
  TRT_NCCL_CHECK(ncclCommInitRank(&comm_, world_, id, rank_));
}

Communicator::~Communicator() {
  if (comm_) (void)ncclCommDestroy(comm_);
  if (scratch_) (void)hipFree(scratch_);
  if (stream_) (void)hipStreamDestroy(stream_);
}

void Communicator::broadcast(void* ptr, size_t bytes, int root,
                             hipStream_t s) {
  TRT_NCCL_CHECK(
      ncclBroadcast(ptr, ptr, bytes, ncclUint8, root, comm_, resolve(s)));
}

void Communicator::all_reduce(void* ptr, size_t count, int dtype, int op,
                              hipStream_t s) {
  TRT_NCCL_CHECK(ncclAllReduce(ptr, ptr, count, nccl_dtype(dtype), red_op(op),
                               comm_, resolve(s)));
}

void Communicator::all_gather(const void* send, void* recv,
                              size_t bytes_per_rank, hipStream_t s) {
  TRT_NCCL_CHECK(
      ncclAllGather(send, recv, bytes_per_rank, ncclUint8, comm_, resolve(s)));
}

void Communicator::reduce_scatter(const void* send, void* recv,
                                  size_t count_per_rank, int dtype, int op,
                                  hipStream_t s) {
  TRT_NCCL_CHECK(ncclReduceScatter(send, recv, count_per_rank,
                                   nccl_dtype(dtype), red_op(op), comm_,
                                   resolve(s)));
}

void Communicator::send(const void* ptr, size_t bytes, int peer,
                        hipStream_t s) {
  TRT_NCCL_CHECK(ncclSend(ptr, bytes, ncclUint8, peer, comm_, resolve(s)));
}

void Communicator::recv(void* ptr, size_t bytes, int peer, hipStream_t s) {
  TRT_NCCL_CHECK(ncclRecv(ptr, bytes, ncclUint8, peer, comm_, resolve(s)));
}

void Communicator::barrier() {
  // all-reduce of one float on the comm stream + host sync: every rank
  // blocks until all ranks arrived (the reference's MPI_Barrier role).
  TRT_HIP_CHECK(hipSetDevice(device_));
  float one = 1.0f;
  TRT_HIP_CHECK(hipMemcpyAsync(scratch_, &one, sizeof(float),
                               hipMemcpyHostToDevice, stream_));
  TRT_NCCL_CHECK(
      ncclAllReduce(scratch_, scratch_, 1, ncclFloat32, ncclSum, comm_,
                    stream_));
  TRT_HIP_CHECK(hipStreamSynchronize(stream_));
}

double Communicator::all_reduce_scalar(double v, int op) {
  TRT_HIP_CHECK(hipSetDevice(device_));
  TRT_HIP_CHECK(hipMemcpyAsync(scratch_, &v, sizeof(double),
                               hipMemcpyHostToDevice, stream_));
  TRT_NCCL_CHECK(ncclAllReduce(scratch_, scratch_, 1, ncclFloat64, red_op(op),
                               comm_, stream_));
  double out = 0.0;
  TRT_HIP_CHECK(hipMemcpyAsync(&out, scratch_, sizeof(double),
                               hipMemcpyDeviceToHost, stream_));
  TRT_HIP_CHECK(hipStreamSynchronize(stream_));
  return out;
}

void Communicator::stream_synchronize() {
  TRT_HIP_CHECK(hipStreamSynchronize(stream_));
}

}  // namespace trtlab
