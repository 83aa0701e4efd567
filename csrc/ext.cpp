// trtlab_amd — pybind11 bindings for the native MI355X runtime + kernels.
// The compute path is hand-written HIP; Python orchestrates. Tensors cross
// the boundary as raw device pointers (torch .data_ptr() or our own arenas).
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <chrono>
#include <thread>

#include "kernels/launchers.h"
#include "runtime/comm.h"
#include "runtime/runtime.h"

namespace py = pybind11;
using namespace trtlab;

namespace {

hipStream_t as_stream(uintptr_t s) { return (hipStream_t)s; }

OpDesc op_from_dict(const py::dict& d) {
  OpDesc o;
  auto gi = [&](const char* k, int def) -> int {
    return d.contains(k) ? d[k].cast<int>() : def;
  };
  auto gl = [&](const char* k, int64_t def) -> int64_t {
    return d.contains(k) ? d[k].cast<int64_t>() : def;
  };
  auto gf = [&](const char* k, float def) -> float {
    return d.contains(k) ? d[k].cast<float>() : def;
  };
  o.kind = gi("kind", 0);
  o.dtype = gi("dtype", 0);
  o.epi = gi("epi", 0);
  o.in_off = gl("in_off", -1);
  o.in2_off = gl("in2_off", -1);
  o.out_off = gl("out_off", -1);
  o.out2_off = gl("out2_off", -1);
  o.out3_off = gl("out3_off", -1);
  o.w_off = gl("w_off", -1);
  o.w2_off = gl("w2_off", -1);
  o.scale_off = gl("scale_off", -1);
  o.bias_off = gl("bias_off", -1);
  o.M = gi("M", 0); o.N = gi("N", 0); o.K = gi("K", 0);
  o.Nb = gi("Nb", 0); o.H = gi("H", 0); o.W = gi("W", 0); o.C = gi("C", 0);
  o.Cout = gi("Cout", 0); o.KH = gi("KH", 0); o.KW = gi("KW", 0);
  o.sh = gi("sh", 1); o.sw = gi("sw", 1); o.ph = gi("ph", 0); o.pw = gi("pw", 0);
  o.HW = gi("HW", 0);
  o.eps = gf("eps", 1e-5f);
  o.n_elems = gl("n_elems", 0);
  o.B = gi("B", 0); o.S = gi("S", 0); o.NH = gi("NH", 0); o.HD = gi("HD", 0);
  o.att_scale = gf("att_scale", 1.0f);
  o.causal = gi("causal", 0);
  o.res_scale = gf("res_scale", 1.0f);
  o.q_scale = gf("q_scale", 1.0f);
  o.tile = gi("tile", 0);
  o.fork = gi("fork", 0);
  o.join = gi("join", 0);
  return o;
}

// Lazily-grown scratch for the raw ops.* test path (serial use only; the
// engine allocates per-context scratch for concurrent replay). Grow-only:
// a captured hipGraph (DecodeSession) bakes the scratch address into its
// nodes, so an old allocation must stay live for the process lifetime —
// regrow allocates a NEW slab and retires the old one without freeing it.
void* test_scratch(size_t need) {
  static std::vector<void*> retired;  // kept live: graphs may reference them
  static void* p = nullptr;
  static size_t cap = 0;
  if (need > cap) {
    if (p) retired.push_back(p);
    TRT_HIP_CHECK(hipMalloc(&p, need));
    cap = need;
  }
  return need ? p : nullptr;
}

// ---- DLPack interop (reference core/types.h:83 DLPack dtypes) ----
// Minimal DLPack ABI structs (the stable v0.x layout torch consumes);
// capsules BORROW our device memory (the owning DeviceBuffer/arena must
// outlive any tensor created from the capsule).
struct DLDevice_ {
  int32_t device_type;
  int32_t device_id;
};
struct DLDataType_ {
  uint8_t code;
  uint8_t bits;
  uint16_t lanes;
};
struct DLTensor_ {
  void* data;
  DLDevice_ device;
  int32_t ndim;
  DLDataType_ dtype;
  int64_t* shape;
  int64_t* strides;
  uint64_t byte_offset;
};
struct DLManagedTensor_ {
  DLTensor_ dl_tensor;
  void* manager_ctx;
  void (*deleter)(DLManagedTensor_*);
};
constexpr int kDLROCM = 10;

struct DLHolder {
  DLManagedTensor_ mt;
  std::vector<int64_t> shape;
};

py::capsule make_dlpack(uintptr_t ptr, std::vector<int64_t> shape,
                        int dtype_code, int bits, int device) {
  auto* h = new DLHolder();
  h->shape = std::move(shape);
  h->mt.dl_tensor.data = (void*)ptr;
  h->mt.dl_tensor.device = {kDLROCM, device};
  h->mt.dl_tensor.ndim = (int32_t)h->shape.size();
  h->mt.dl_tensor.dtype = {(uint8_t)dtype_code, (uint8_t)bits, 1};
  h->mt.dl_tensor.shape = h->shape.data();
  h->mt.dl_tensor.strides = nullptr;  // compact row-major
  h->mt.dl_tensor.byte_offset = 0;
  h->mt.manager_ctx = h;
  h->mt.deleter = [](DLManagedTensor_* mt) {
    delete (DLHolder*)mt->manager_ctx;  // memory itself is borrowed
  };
  return py::capsule(&h->mt, "dltensor", [](PyObject* cap) {
    // unconsumed capsule: torch renames it to "used_dltensor" when it
    // takes ownership, so only delete if still named "dltensor"
    if (PyCapsule_IsValid(cap, "dltensor")) {
      auto* mt = (DLManagedTensor_*)PyCapsule_GetPointer(cap, "dltensor");
      if (mt && mt->deleter) mt->deleter(mt);
    }
  });
}

}  // namespace

PYBIND11_MODULE(_C, m) {
  m.doc() = "trtlab_amd native runtime (gfx950/CDNA4)";

  // ------------------------------------------------------------- memory
  auto mem = m.def_submodule("memory");
  mem.def("device_malloc", [](size_t b, int dev) {
    return (uintptr_t)device_malloc(b, dev);
  });
  mem.def("device_free", [](uintptr_t p, size_t b) { device_free((void*)p, b); });
  mem.def("pinned_malloc", [](size_t b) { return (uintptr_t)pinned_malloc(b); });
  mem.def("pinned_free", [](uintptr_t p, size_t b) { pinned_free((void*)p, b); });
  mem.def("device_bytes_in_use", &device_bytes_in_use);
  mem.def("pinned_bytes_in_use", &pinned_bytes_in_use);
  mem.def("huge_malloc", [](size_t b, bool pin) {
    bool hugetlb = false;
    void* p = huge_malloc(b, pin, &hugetlb);
    return py::make_tuple((uintptr_t)p, hugetlb);
  });
  mem.def("huge_free",
          [](uintptr_t p, size_t b, bool pin) { huge_free((void*)p, b, pin); });
  mem.def("huge_bytes_in_use", &huge_bytes_in_use);
  mem.def("host_view", [](uintptr_t p, size_t bytes) {
    return py::memoryview::from_memory((void*)p, bytes);
  });
  mem.def("memcpy_h2d", [](uintptr_t dst, py::buffer src, size_t bytes) {
    py::buffer_info info = src.request();
    TRT_HIP_CHECK(hipMemcpy((void*)dst, info.ptr, bytes, hipMemcpyHostToDevice));
  });
  mem.def("memcpy_d2h", [](py::buffer dst, uintptr_t src, size_t bytes) {
    py::buffer_info info = dst.request();
    TRT_HIP_CHECK(hipMemcpy(info.ptr, (void*)src, bytes, hipMemcpyDeviceToHost));
  });
  mem.def("memcpy_d2d", [](uintptr_t dst, uintptr_t src, size_t bytes) {
    TRT_HIP_CHECK(hipMemcpy((void*)dst, (void*)src, bytes, hipMemcpyDeviceToDevice));
  });
  mem.def("memset_d", [](uintptr_t p, int v, size_t bytes) {
    TRT_HIP_CHECK(hipMemset((void*)p, v, bytes));
  });

  // DLPack export: dtype_code 2 = float (fp16 bits=16, fp32 bits=32),
  // 0 = int, 1 = uint, 4 = bfloat. torch.from_dlpack(capsule) gives a
  // ZERO-COPY tensor view of our device memory.
  mem.def("to_dlpack",
          [](uintptr_t ptr, std::vector<int64_t> shape, int dtype_code,
             int bits, int device) {
            return make_dlpack(ptr, std::move(shape), dtype_code, bits,
                               device);
          },
          py::arg("ptr"), py::arg("shape"), py::arg("dtype_code") = 2,
          py::arg("bits") = 16, py::arg("device") = 0);

  // Growing best-fit device allocator (reference bfit_allocator +
  // growing block_arena) for multi-model serving / dynamic shapes.
  py::class_<DeviceArena>(mem, "DeviceArena")
      .def(py::init<int, size_t, size_t, size_t>(), py::arg("device") = 0,
           py::arg("initial_bytes") = 0, py::arg("max_bytes") = 0,
           py::arg("growth_bytes") = 0)
      .def("allocate",
           [](DeviceArena& a, size_t bytes, size_t align) {
             return (uintptr_t)a.allocate(bytes, align);
           },
           py::arg("bytes"), py::arg("align") = 256)
      .def("deallocate",
           [](DeviceArena& a, uintptr_t p) { a.deallocate((void*)p); })
      .def("stats", [](DeviceArena& a) {
        auto s = a.stats();
        py::dict d;
        d["capacity"] = s.capacity;
        d["in_use"] = s.in_use;
        d["high_water"] = s.high_water;
        d["largest_free"] = s.largest_free;
        d["free_nodes"] = s.free_nodes;
        d["live_allocs"] = s.live_allocs;
        py::list h;
        for (int i = 0; i < 48; ++i) h.append(s.histogram[i]);
        d["histogram"] = h;
        return d;
      });

  // hybrid futex mutex/cv stress (host-only test hook): N threads bang a
  // shared counter under HybridMutex; returns (counter, ms). A lost
  // update proves a broken lock, so the test asserts exactly T*I.
  mem.def("hybrid_mutex_stress", [](int threads, int iters) {
    HybridMutex mu;
    HybridCondition cv;
    long counter = 0;
    bool go = false;
    std::vector<std::thread> ts;
    auto t0 = std::chrono::steady_clock::now();
    for (int t = 0; t < threads; ++t)
      ts.emplace_back([&] {
        {
          HybridLock g(mu);
          while (!go) cv.wait(mu);
        }
        for (int i = 0; i < iters; ++i) {
          HybridLock g(mu);
          ++counter;
        }
      });
    {
      HybridLock g(mu);
      go = true;
      cv.notify_all();
    }
    for (auto& th : ts) th.join();
    double ms = std::chrono::duration<double, std::milli>(
                    std::chrono::steady_clock::now() - t0)
                    .count();
    return py::make_tuple(counter, ms);
  }, py::arg("threads") = 8, py::arg("iters") = 100000);

  py::class_<BlockPool>(mem, "BlockPool")
      .def(py::init<size_t, int, int>(), py::arg("block_bytes"),
           py::arg("count"), py::arg("device") = 0)
      .def("acquire", [](BlockPool& p) { return (uintptr_t)p.acquire(); })
      .def("release", [](BlockPool& p, uintptr_t b) { p.release((void*)b); })
      .def("available", &BlockPool::available)
      .def_property_readonly("block_bytes", &BlockPool::block_bytes)
      .def_property_readonly("total", &BlockPool::total);

  // ---------------------------------------------------------------- hip
  auto hip = m.def_submodule("hip");
  hip.def("last_error", [] {
    hipError_t e = hipGetLastError();
    return std::string(hipGetErrorString(e));
  });
  hip.def("device_count", [] {
    int n = 0;
    hipError_t e = hipGetDeviceCount(&n);
    return e == hipSuccess ? n : 0;
  });
  hip.def("set_device", [](int d) { TRT_HIP_CHECK(hipSetDevice(d)); });
  hip.def("device_synchronize", [] {
    py::gil_scoped_release rel;
    TRT_HIP_CHECK(hipDeviceSynchronize());
  });
  hip.def("stream_create", [] {
    hipStream_t s;
    TRT_HIP_CHECK(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
    return (uintptr_t)s;
  });
  hip.def("stream_destroy", [](uintptr_t s) { hipStreamDestroy(as_stream(s)); });
  hip.def("stream_synchronize", [](uintptr_t s) {
    py::gil_scoped_release rel;
    TRT_HIP_CHECK(hipStreamSynchronize(as_stream(s)));
  });
  // generic stream capture (DecodeSession and power users build their own
  // replayable graphs from raw ops)
  hip.def("stream_begin_capture", [](uintptr_t s) {
    TRT_HIP_CHECK(hipStreamBeginCapture(as_stream(s),
                                        hipStreamCaptureModeRelaxed));
  });
  hip.def("stream_end_capture", [](uintptr_t s) {
    hipGraph_t g{};
    TRT_HIP_CHECK(hipStreamEndCapture(as_stream(s), &g));
    hipGraphExec_t e{};
    TRT_HIP_CHECK(hipGraphInstantiate(&e, g, nullptr, nullptr, 0));
    TRT_HIP_CHECK(hipGraphDestroy(g));
    return (uintptr_t)e;
  });
  hip.def("graph_launch", [](uintptr_t e, uintptr_t s) {
    TRT_HIP_CHECK(hipGraphLaunch((hipGraphExec_t)e, as_stream(s)));
  });
  hip.def("graph_destroy", [](uintptr_t e) {
    (void)hipGraphExecDestroy((hipGraphExec_t)e);
  });
  hip.def("device_properties", [](int dev) {
    hipDeviceProp_t p;
    TRT_HIP_CHECK(hipGetDeviceProperties(&p, dev));
    py::dict d;
    d["name"] = std::string(p.name);
    d["gcn_arch"] = std::string(p.gcnArchName);
    d["total_mem"] = (int64_t)p.totalGlobalMem;
    d["multi_processor_count"] = p.multiProcessorCount;
    d["lds_per_block"] = (int64_t)p.sharedMemPerBlock;
    d["clock_khz"] = p.clockRate;
    d["warp_size"] = p.warpSize;
    return d;
  });

  // ---------------------------------------------------------------- ops
  // Raw kernel launchers (tests + eager use). Pointers are integers
  // (torch .data_ptr() or memory.device_malloc). stream 0 = default.
  auto ops = m.def_submodule("ops");
  ops.def("gemm_bt",
          [](int dtype, uintptr_t A, uintptr_t B, uintptr_t C, uintptr_t scale,
             uintptr_t bias, uintptr_t residual, int M, int N, int K, int epi,
             uintptr_t stream, bool sync, int tile, float res_scale,
             int64_t lda, int64_t ldb, int64_t ldc) {
            // strided operands: lda/ldb/ldc in ELEMENTS (0 = contiguous
            // [M,K]/[N,K]/[M,N]) — sub-matrix views / transposed outputs
            launch_gemm_bt(dtype, (void*)A, (void*)B, (void*)C, (float*)scale,
                           (float*)bias, (void*)residual, res_scale, M, N, K,
                           lda ? lda : K, ldb ? ldb : K, ldc ? ldc : N, epi,
                           as_stream(stream), tile,
                           test_scratch(gemm_scratch_bytes(M, N, K)));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("A"), py::arg("B"), py::arg("C"),
          py::arg("scale") = 0, py::arg("bias") = 0, py::arg("residual") = 0,
          py::arg("M") = 0, py::arg("N") = 0, py::arg("K") = 0,
          py::arg("epi") = 0, py::arg("stream") = 0, py::arg("sync") = true,
          py::arg("tile") = 0, py::arg("res_scale") = 1.0f,
          py::arg("lda") = 0, py::arg("ldb") = 0, py::arg("ldc") = 0);
  ops.def("conv2d",
          [](int dtype, uintptr_t in, uintptr_t Wt, uintptr_t out,
             uintptr_t scale, uintptr_t bias, uintptr_t residual,
             uintptr_t zero_page, int Nb, int H, int W, int C, int Cout,
             int KH, int KW, int sh, int sw, int ph, int pw, int epi,
             uintptr_t stream, bool sync, int tile, float res_scale) {
            launch_conv2d(dtype, (void*)in, (void*)Wt, (void*)out,
                          (float*)scale, (float*)bias, (void*)residual,
                          (void*)zero_page, Nb, H, W, C, Cout, KH, KW, sh, sw,
                          ph, pw, epi, as_stream(stream), tile,
                          test_scratch(conv_scratch_bytes(
                              Nb, H, W, C, Cout, KH, KW, sh, sw, ph, pw)),
                          res_scale);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("in"), py::arg("Wt"), py::arg("out"),
          py::arg("scale") = 0, py::arg("bias") = 0, py::arg("residual") = 0,
          py::arg("zero_page") = 0, py::arg("Nb") = 1, py::arg("H") = 0,
          py::arg("W") = 0, py::arg("C") = 0, py::arg("Cout") = 0,
          py::arg("KH") = 1, py::arg("KW") = 1, py::arg("sh") = 1,
          py::arg("sw") = 1, py::arg("ph") = 0, py::arg("pw") = 0,
          py::arg("epi") = 0, py::arg("stream") = 0, py::arg("sync") = true,
          py::arg("tile") = 0, py::arg("res_scale") = 1.0f);
  ops.def("bottleneck_tail",
          [](int dtype, uintptr_t in, uintptr_t W1, uintptr_t W2,
             uintptr_t out, uintptr_t s1, uintptr_t b1, uintptr_t s2,
             uintptr_t b2, uintptr_t residual, uintptr_t zero_page, int Nb,
             int H, int W, int Cm, int Co, uintptr_t stream, bool sync) {
            launch_bottleneck_tail(dtype, (void*)in, (void*)W1, (void*)W2,
                                   (void*)out, (float*)s1, (float*)b1,
                                   (float*)s2, (float*)b2, (void*)residual,
                                   (void*)zero_page, Nb, H, W, Cm, Co,
                                   as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("in"), py::arg("W1"), py::arg("W2"),
          py::arg("out"), py::arg("s1"), py::arg("b1"), py::arg("s2"),
          py::arg("b2"), py::arg("residual"), py::arg("zero_page"),
          py::arg("Nb"), py::arg("H"), py::arg("W"), py::arg("Cm"),
          py::arg("Co"), py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("maxpool2d",
          [](int dtype, uintptr_t in, uintptr_t out, int Nb, int H, int W,
             int C, int KH, int KW, int sh, int sw, int ph, int pw,
             uintptr_t stream, bool sync) {
            launch_maxpool2d(dtype, (void*)in, (void*)out, Nb, H, W, C, KH,
                             KW, sh, sw, ph, pw, as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("in"), py::arg("out"), py::arg("Nb"),
          py::arg("H"), py::arg("W"), py::arg("C"), py::arg("KH"),
          py::arg("KW"), py::arg("sh"), py::arg("sw"), py::arg("ph"),
          py::arg("pw"), py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("avgpool2d",
          [](int dtype, uintptr_t in, uintptr_t out, int Nb, int H, int W,
             int C, int KH, int KW, int sh, int sw, int ph, int pw,
             uintptr_t stream, bool sync) {
            launch_avgpool2d(dtype, (void*)in, (void*)out, Nb, H, W, C, KH,
                             KW, sh, sw, ph, pw, as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("in"), py::arg("out"), py::arg("Nb"),
          py::arg("H"), py::arg("W"), py::arg("C"), py::arg("KH"),
          py::arg("KW"), py::arg("sh"), py::arg("sw"), py::arg("ph"),
          py::arg("pw"), py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("gavgpool",
          [](int dtype, uintptr_t in, uintptr_t out, int Nb, int HW, int C,
             uintptr_t stream, bool sync) {
            launch_gavgpool(dtype, (void*)in, (void*)out, Nb, HW, C,
                            as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("in"), py::arg("out"), py::arg("Nb"),
          py::arg("HW"), py::arg("C"), py::arg("stream") = 0,
          py::arg("sync") = true);
  ops.def("softmax_rows",
          [](int dtype, uintptr_t in, uintptr_t out, int M, int N,
             uintptr_t stream, bool sync) {
            launch_softmax_rows(dtype, (void*)in, (void*)out, M, N, N,
                                as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("in"), py::arg("out"), py::arg("M"),
          py::arg("N"), py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("layernorm",
          [](int dtype, uintptr_t in, uintptr_t gamma, uintptr_t beta,
             uintptr_t out, int M, int N, float eps, uintptr_t stream,
             bool sync) {
            launch_layernorm(dtype, (void*)in, (float*)gamma, (float*)beta,
                             (void*)out, M, N, N, eps, as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("in"), py::arg("gamma"), py::arg("beta"),
          py::arg("out"), py::arg("M"), py::arg("N"), py::arg("eps") = 1e-5f,
          py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("add_layernorm",
          [](int dtype, uintptr_t x, uintptr_t res, uintptr_t gamma,
             uintptr_t beta, uintptr_t out, uintptr_t sum_out, int M, int N,
             float eps, uintptr_t stream, bool sync) {
            launch_add_layernorm(dtype, (void*)x, (void*)res, (float*)gamma,
                                 (float*)beta, (void*)out, (void*)sum_out, M,
                                 N, N, eps, as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("x"), py::arg("res"), py::arg("gamma"),
          py::arg("beta"), py::arg("out"), py::arg("sum_out") = 0,
          py::arg("M") = 0, py::arg("N") = 0, py::arg("eps") = 1e-5f,
          py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("elementwise",
          [](int dtype, int op, uintptr_t a, uintptr_t b, uintptr_t out,
             int64_t n, uintptr_t stream, bool sync) {
            launch_elementwise(dtype, op, (void*)a, (void*)b, (void*)out, n,
                               as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("op"), py::arg("a"), py::arg("b") = 0,
          py::arg("out") = 0, py::arg("n") = 0, py::arg("stream") = 0,
          py::arg("sync") = true);
  ops.def("rmsnorm",
          [](int dtype, uintptr_t in, uintptr_t gamma, uintptr_t out, int M,
             int N, float eps, uintptr_t stream, bool sync) {
            launch_rmsnorm(dtype, (void*)in, (float*)gamma, (void*)out, M,
                           N, N, eps, as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("in"), py::arg("gamma"), py::arg("out"),
          py::arg("M"), py::arg("N"), py::arg("eps") = 1e-5f,
          py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("add_rmsnorm",
          [](int dtype, uintptr_t x, uintptr_t res, uintptr_t gamma,
             uintptr_t out, uintptr_t sum_out, int M, int N, float eps,
             uintptr_t stream, bool sync) {
            launch_add_rmsnorm(dtype, (void*)x, (void*)res, (float*)gamma,
                               (void*)out, (void*)sum_out, M, N, N, eps,
                               as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("x"), py::arg("res"), py::arg("gamma"),
          py::arg("out"), py::arg("sum_out") = 0, py::arg("M") = 0,
          py::arg("N") = 0, py::arg("eps") = 1e-5f, py::arg("stream") = 0,
          py::arg("sync") = true);
  ops.def("silu_mul",
          [](int dtype, uintptr_t a, uintptr_t b, uintptr_t out, int64_t n,
             uintptr_t stream, bool sync) {
            launch_silu_mul(dtype, (void*)a, (void*)b, (void*)out, n,
                            as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("a"), py::arg("b"), py::arg("out"),
          py::arg("n"), py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("rope",
          [](int dtype, uintptr_t qkv, uintptr_t pos, int M, int S, int H,
             int D, float theta, uintptr_t stream, bool sync, int chunk) {
            launch_rope(dtype, (void*)qkv, (void*)pos, M, S, H, D, theta,
                        as_stream(stream), chunk);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("qkv"), py::arg("pos") = 0,
          py::arg("M") = 0, py::arg("S") = 0, py::arg("H") = 0,
          py::arg("D") = 0, py::arg("theta") = 10000.0f,
          py::arg("stream") = 0, py::arg("sync") = true,
          py::arg("chunk") = 0);
  ops.def("argmax_rows",
          [](uintptr_t x, uintptr_t out, int M, int V, uintptr_t stream,
             bool sync) {
            launch_argmax_rows((void*)x, (void*)out, M, V,
                               as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("x"), py::arg("out"), py::arg("M"), py::arg("V"),
          py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("gumbel_argmax_rows",
          [](uintptr_t x, uintptr_t out, uintptr_t temps, uintptr_t seeds,
             uintptr_t pos, int M, int V, uintptr_t stream, bool sync) {
            launch_gumbel_argmax_rows((void*)x, (void*)out, (void*)temps,
                                      (void*)seeds, (void*)pos, M, V,
                                      as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("x"), py::arg("out"), py::arg("temps") = 0,
          py::arg("seeds") = 0, py::arg("pos") = 0, py::arg("M") = 0,
          py::arg("V") = 0, py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("channel_affine",
          [](int dtype, uintptr_t x, uintptr_t out, uintptr_t s,
             uintptr_t b, int64_t M, int C, bool relu, uintptr_t stream,
             bool sync) {
            launch_channel_affine(dtype, (void*)x, (void*)out, (float*)s,
                                  (float*)b, M, C, relu,
                                  as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("x"), py::arg("out"), py::arg("s"),
          py::arg("b"), py::arg("M"), py::arg("C"), py::arg("relu") = false,
          py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("clip",
          [](int dtype, uintptr_t in, uintptr_t out, int64_t n, float mn,
             float mx, uintptr_t stream, bool sync) {
            launch_clip(dtype, (void*)in, (void*)out, n, mn, mx,
                        as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("in"), py::arg("out"), py::arg("n"),
          py::arg("mn"), py::arg("mx"), py::arg("stream") = 0,
          py::arg("sync") = true);
  ops.def("transpose2d",
          [](int dtype, uintptr_t in, uintptr_t out, int M, int N,
             uintptr_t stream, bool sync) {
            launch_transpose2d(dtype, (void*)in, (void*)out, M, N,
                               as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("in"), py::arg("out"), py::arg("M"),
          py::arg("N"), py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("copy2d",
          [](int dtype, uintptr_t src, uintptr_t dst, int64_t M, int C,
             int ldd, int coff, uintptr_t stream, bool sync) {
            launch_copy2d(dtype, (void*)src, (void*)dst, M, C, ldd, coff,
                          as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("src"), py::arg("dst"), py::arg("M"),
          py::arg("C"), py::arg("ldd"), py::arg("coff") = 0,
          py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("channel_pad",
          [](int dtype, uintptr_t in, uintptr_t out, int64_t M, int Cin,
             int Cpad, uintptr_t stream, bool sync) {
            launch_channel_pad(dtype, (void*)in, (void*)out, M, Cin, Cpad,
                               as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("in"), py::arg("out"), py::arg("M"),
          py::arg("Cin"), py::arg("Cpad"), py::arg("stream") = 0,
          py::arg("sync") = true);
  ops.def("quantize",
          [](uintptr_t in, uintptr_t out, int64_t n, float scale,
             uintptr_t stream, bool sync, int fmt) {
            launch_quantize((void*)in, (void*)out, n, scale, as_stream(stream),
                            fmt);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("in"), py::arg("out"), py::arg("n"), py::arg("scale"),
          py::arg("stream") = 0, py::arg("sync") = true, py::arg("fmt") = 0);
  ops.def("dequant",
          [](uintptr_t in, uintptr_t out, int64_t n, float scale,
             uintptr_t stream, bool sync, int fmt) {
            launch_dequant((void*)in, (void*)out, n, scale, as_stream(stream),
                           fmt);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("in"), py::arg("out"), py::arg("n"), py::arg("scale"),
          py::arg("stream") = 0, py::arg("sync") = true, py::arg("fmt") = 0);
  ops.def("embedding",
          [](int dtype, uintptr_t ids, uintptr_t tok, uintptr_t pos,
             uintptr_t seg, uintptr_t segids, uintptr_t out, int M, int S,
             int H, uintptr_t stream, bool sync) {
            launch_embedding(dtype, (void*)ids, (void*)tok, (void*)pos,
                             (void*)seg, (void*)segids, (void*)out, M, S, H,
                             as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("ids"), py::arg("tok"), py::arg("pos"),
          py::arg("seg") = 0, py::arg("segids") = 0, py::arg("out") = 0,
          py::arg("M") = 0, py::arg("S") = 0, py::arg("H") = 0,
          py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("mx_probe",
          [](uintptr_t A, uintptr_t B, uintptr_t Sa, uintptr_t Sb,
             uintptr_t D) {
            launch_mx_probe((void*)A, (void*)B, (void*)Sa, (void*)Sb,
                            (void*)D, 0);
            TRT_HIP_CHECK(hipStreamSynchronize(0));
          });
  ops.def("mx_frag_dump", [](uintptr_t A, uintptr_t out, int K) {
    launch_mx_frag_dump((void*)A, (void*)out, K, 0);
    TRT_HIP_CHECK(hipStreamSynchronize(0));
  });
  ops.def("quantize_mxfp4",
          [](uintptr_t x, uintptr_t codes, uintptr_t scales, int64_t m,
             int64_t k, uintptr_t stream, bool sync) {
            launch_quantize_mxfp4((void*)x, (void*)codes, (void*)scales, m, k,
                                  as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("x"), py::arg("codes"), py::arg("scales"), py::arg("m"),
          py::arg("k"), py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("gemm_mxfp4",
          [](uintptr_t A, uintptr_t B, uintptr_t Sa, uintptr_t Sb,
             uintptr_t C, int M, int N, int K, uintptr_t stream, bool sync) {
            launch_gemm_mxfp4((void*)A, (void*)B, (void*)Sa, (void*)Sb,
                              (void*)C, M, N, K, as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("A"), py::arg("B"), py::arg("Sa"), py::arg("Sb"),
          py::arg("C"), py::arg("M"), py::arg("N"), py::arg("K"),
          py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("mx4_probe",
          [](uintptr_t A, uintptr_t B, uintptr_t Sa, uintptr_t Sb,
             uintptr_t D) {
            launch_mx4_probe((void*)A, (void*)B, (void*)Sa, (void*)Sb,
                             (void*)D, 0);
            TRT_HIP_CHECK(hipStreamSynchronize(0));
          });
  ops.def("gemm_mxfp8",
          [](uintptr_t A, uintptr_t B, uintptr_t Sa, uintptr_t Sb,
             uintptr_t C, int M, int N, int K, uintptr_t stream, bool sync) {
            launch_gemm_mxfp8((void*)A, (void*)B, (void*)Sa, (void*)Sb,
                              (void*)C, M, N, K, as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("A"), py::arg("B"), py::arg("Sa"), py::arg("Sb"),
          py::arg("C"), py::arg("M"), py::arg("N"), py::arg("K"),
          py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("kv_append",
          [](uintptr_t qkv, uintptr_t kc, uintptr_t vc, uintptr_t pos, int B,
             int H, int smax, uintptr_t stream, bool sync, int D) {
            launch_kv_append((void*)qkv, (void*)kc, (void*)vc, (void*)pos, B,
                             H, smax, as_stream(stream), D);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("qkv"), py::arg("kcache"), py::arg("vcache"),
          py::arg("pos"), py::arg("B"), py::arg("H"), py::arg("smax"),
          py::arg("stream") = 0, py::arg("sync") = true,
          py::arg("D") = 64);
  ops.def("kv_append_range",
          [](uintptr_t qkv, uintptr_t kc, uintptr_t vc, int B, int H, int P,
             int smax, uintptr_t stream, bool sync, int D) {
            launch_kv_append_range((void*)qkv, (void*)kc, (void*)vc, B, H, P,
                                   smax, as_stream(stream), D);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("qkv"), py::arg("kcache"), py::arg("vcache"), py::arg("B"),
          py::arg("H"), py::arg("P"), py::arg("smax"), py::arg("stream") = 0,
          py::arg("sync") = true, py::arg("D") = 64);
  ops.def("kv_append_chunk_paged",
          [](uintptr_t qkv, uintptr_t kpool, uintptr_t vpool,
             uintptr_t table, uintptr_t pos, int B, int H, int K,
             int max_pages, uintptr_t stream, bool sync, int D) {
            launch_kv_append_chunk_paged((void*)qkv, (void*)kpool,
                                         (void*)vpool, (void*)table,
                                         (void*)pos, B, H, K, max_pages,
                                         as_stream(stream), D);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("qkv"), py::arg("kpool"), py::arg("vpool"),
          py::arg("table"), py::arg("pos"), py::arg("B"), py::arg("H"),
          py::arg("K"), py::arg("max_pages"), py::arg("stream") = 0,
          py::arg("sync") = true, py::arg("D") = 64);
  ops.def("kv_append_range_paged",
          [](uintptr_t qkv, uintptr_t kpool, uintptr_t vpool,
             uintptr_t table, int B, int H, int P, int max_pages,
             uintptr_t stream, bool sync, int D) {
            launch_kv_append_range_paged((void*)qkv, (void*)kpool,
                                         (void*)vpool, (void*)table, B, H,
                                         P, max_pages, as_stream(stream),
                                         D);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("qkv"), py::arg("kpool"), py::arg("vpool"),
          py::arg("table"), py::arg("B"), py::arg("H"), py::arg("P"),
          py::arg("max_pages"), py::arg("stream") = 0,
          py::arg("sync") = true, py::arg("D") = 64);
  ops.def("chunk_attention_paged",
          [](uintptr_t qkv, uintptr_t kpool, uintptr_t vpool, uintptr_t out,
             uintptr_t table, uintptr_t pos, int B, int H, int K,
             int max_pages, float scale, uintptr_t stream, bool sync,
             int D) {
            launch_chunk_attention_paged((void*)qkv, (void*)kpool,
                                         (void*)vpool, (void*)out,
                                         (void*)table, (void*)pos, B, H, K,
                                         max_pages, scale,
                                         as_stream(stream), D);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("qkv"), py::arg("kpool"), py::arg("vpool"),
          py::arg("out"), py::arg("table"), py::arg("pos"), py::arg("B"),
          py::arg("H"), py::arg("K"), py::arg("max_pages"),
          py::arg("scale"), py::arg("stream") = 0, py::arg("sync") = true,
          py::arg("D") = 64);
  ops.def("kv_append_paged",
          [](uintptr_t qkv, uintptr_t kp, uintptr_t vp, uintptr_t table,
             uintptr_t pos, int B, int H, int max_pages, uintptr_t stream,
             bool sync, int D) {
            launch_kv_append_paged((void*)qkv, (void*)kp, (void*)vp,
                                   (void*)table, (void*)pos, B, H,
                                   max_pages, as_stream(stream), D);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("qkv"), py::arg("kpool"), py::arg("vpool"),
          py::arg("table"), py::arg("pos"), py::arg("B"), py::arg("H"),
          py::arg("max_pages"), py::arg("stream") = 0,
          py::arg("sync") = true, py::arg("D") = 64);
  ops.def("decode_attention_paged",
          [](uintptr_t qkv, uintptr_t kp, uintptr_t vp, uintptr_t out,
             uintptr_t table, uintptr_t pos, int B, int H, int max_pages,
             float scale, uintptr_t stream, bool sync, int D) {
            launch_decode_attention_paged((void*)qkv, (void*)kp, (void*)vp,
                                          (void*)out, (void*)table,
                                          (void*)pos, B, H, max_pages,
                                          scale, as_stream(stream), D);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("qkv"), py::arg("kpool"), py::arg("vpool"), py::arg("out"),
          py::arg("table"), py::arg("pos"), py::arg("B"), py::arg("H"),
          py::arg("max_pages"), py::arg("scale"), py::arg("stream") = 0,
          py::arg("sync") = true, py::arg("D") = 64);
  ops.def("decode_attention",
          [](uintptr_t qkv, uintptr_t kc, uintptr_t vc, uintptr_t out,
             uintptr_t pos, int B, int H, int smax, float scale,
             uintptr_t stream, bool sync, int D) {
            launch_decode_attention((void*)qkv, (void*)kc, (void*)vc,
                                    (void*)out, (void*)pos, B, H, smax,
                                    scale, as_stream(stream), D);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("qkv"), py::arg("kcache"), py::arg("vcache"),
          py::arg("out"), py::arg("pos"), py::arg("B"), py::arg("H"),
          py::arg("smax"), py::arg("scale"), py::arg("stream") = 0,
          py::arg("sync") = true, py::arg("D") = 64);
  ops.def("decode_embed",
          [](uintptr_t ids, uintptr_t tok, uintptr_t pe, uintptr_t out,
             uintptr_t pos, int B, int hidden, uintptr_t stream, bool sync) {
            launch_decode_embed((void*)ids, (void*)tok, (void*)pe, (void*)out,
                                (void*)pos, B, hidden, as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("ids"), py::arg("tok"), py::arg("posemb"), py::arg("out"),
          py::arg("pos"), py::arg("B"), py::arg("hidden"),
          py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("decode_gemm_fused",
          [](int pro, int epi, uintptr_t x, uintptr_t r, uintptr_t h_out,
             uintptr_t gamma, uintptr_t beta, uintptr_t Bw, uintptr_t bias,
             uintptr_t C, uintptr_t ids, uintptr_t tok, uintptr_t posemb,
             uintptr_t pos, uintptr_t kcache, uintptr_t vcache, int M,
             int N, int K, int heads, int smax, float eps, uintptr_t stream,
             bool sync) {
            launch_decode_gemm_fused(
                pro, epi, (const void*)x, (const void*)r, (void*)h_out,
                (const float*)gamma, (const float*)beta, (const void*)Bw,
                (const float*)bias, (void*)C, (const void*)ids,
                (const void*)tok, (const void*)posemb, (const void*)pos,
                (void*)kcache, (void*)vcache, M, N, K, heads, smax, eps,
                as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("pro"), py::arg("epi"), py::arg("x") = 0, py::arg("r") = 0,
          py::arg("h_out") = 0, py::arg("gamma") = 0, py::arg("beta") = 0,
          py::arg("B") = 0, py::arg("bias") = 0, py::arg("C") = 0,
          py::arg("ids") = 0, py::arg("tok") = 0, py::arg("posemb") = 0,
          py::arg("pos") = 0, py::arg("kcache") = 0, py::arg("vcache") = 0,
          py::arg("M") = 0, py::arg("N") = 0, py::arg("K") = 0,
          py::arg("heads") = 0, py::arg("smax") = 0, py::arg("eps") = 1e-5f,
          py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("kv_append_chunk",
          [](uintptr_t qkv, uintptr_t kc, uintptr_t vc, uintptr_t pos, int B,
             int H, int K, int smax, uintptr_t stream, bool sync, int D) {
            launch_kv_append_chunk((void*)qkv, (void*)kc, (void*)vc,
                                   (void*)pos, B, H, K, smax,
                                   as_stream(stream), D);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("qkv"), py::arg("kcache"), py::arg("vcache"),
          py::arg("pos"), py::arg("B"), py::arg("H"), py::arg("K"),
          py::arg("smax"), py::arg("stream") = 0, py::arg("sync") = true,
          py::arg("D") = 64);
  ops.def("chunk_attention",
          [](uintptr_t qkv, uintptr_t kc, uintptr_t vc, uintptr_t out,
             uintptr_t pos, int B, int H, int K, int smax, float scale,
             uintptr_t stream, bool sync, int D) {
            launch_chunk_attention((void*)qkv, (void*)kc, (void*)vc,
                                   (void*)out, (void*)pos, B, H, K, smax,
                                   scale, as_stream(stream), D);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("qkv"), py::arg("kcache"), py::arg("vcache"),
          py::arg("out"), py::arg("pos"), py::arg("B"), py::arg("H"),
          py::arg("K"), py::arg("smax"), py::arg("scale"),
          py::arg("stream") = 0, py::arg("sync") = true,
          py::arg("D") = 64);
  ops.def("chunk_embed",
          [](uintptr_t ids, uintptr_t tok, uintptr_t pe, uintptr_t out,
             uintptr_t pos, int B, int K, int smax, int hidden,
             uintptr_t stream, bool sync) {
            launch_chunk_embed((void*)ids, (void*)tok, (void*)pe, (void*)out,
                               (void*)pos, B, K, smax, hidden,
                               as_stream(stream));
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("ids"), py::arg("tok"), py::arg("posemb"), py::arg("out"),
          py::arg("pos"), py::arg("B"), py::arg("K"), py::arg("smax"),
          py::arg("hidden"), py::arg("stream") = 0, py::arg("sync") = true);
  ops.def("advance_pos", [](uintptr_t pos, int B, int smax, uintptr_t stream,
                            bool sync) {
    launch_advance_pos((void*)pos, B, smax, as_stream(stream));
    if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
  }, py::arg("pos"), py::arg("B"), py::arg("smax"), py::arg("stream") = 0,
     py::arg("sync") = true);
  ops.def("attention",
          [](int dtype, uintptr_t qkv, uintptr_t out, int B, int S, int H,
             int D, float scale, uintptr_t stream, bool sync, int causal,
             uintptr_t seqlens) {
            launch_attention(dtype, (void*)qkv, (void*)out, B, S, H, D, scale,
                             as_stream(stream), -1, 1.0f, (void*)seqlens,
                             causal);
            if (sync) TRT_HIP_CHECK(hipStreamSynchronize(as_stream(stream)));
          },
          py::arg("dtype"), py::arg("qkv"), py::arg("out"), py::arg("B"),
          py::arg("S"), py::arg("H"), py::arg("D"), py::arg("scale"),
          py::arg("stream") = 0, py::arg("sync") = true,
          py::arg("causal") = 0, py::arg("seqlens") = 0);

  // --------------------------------------------------------------- comm
  // Owned RCCL collective layer (no torch.distributed in the data path).
  auto comm = m.def_submodule("comm");
  comm.def("unique_id", [] { return py::bytes(Communicator::unique_id()); });
  py::class_<Communicator>(comm, "Communicator")
      .def(py::init([](int rank, int world, py::bytes uid, int device) {
             // clique connect can block for seconds — release the GIL so
             // multi-rank-per-process tests don't deadlock
             std::string u = uid;
             py::gil_scoped_release rel;
             return new Communicator(rank, world, u, device);
           }),
           py::arg("rank"), py::arg("world"), py::arg("uid"),
           py::arg("device") = 0)
      .def_property_readonly("rank", &Communicator::rank)
      .def_property_readonly("world", &Communicator::world)
      .def_property_readonly("device", &Communicator::device)
      .def("broadcast",
           [](Communicator& c, uintptr_t ptr, size_t bytes, int root,
              uintptr_t stream) {
             py::gil_scoped_release rel;
             c.broadcast((void*)ptr, bytes, root, as_stream(stream));
           },
           py::arg("ptr"), py::arg("bytes"), py::arg("root") = 0,
           py::arg("stream") = 0)
      .def("all_reduce",
           [](Communicator& c, uintptr_t ptr, size_t count, int dtype, int op,
              uintptr_t stream) {
             py::gil_scoped_release rel;
             c.all_reduce((void*)ptr, count, dtype, op, as_stream(stream));
           },
           py::arg("ptr"), py::arg("count"), py::arg("dtype") = 3,
           py::arg("op") = 0, py::arg("stream") = 0)
      .def("all_gather",
           [](Communicator& c, uintptr_t send, uintptr_t recv,
              size_t bytes_per_rank, uintptr_t stream) {
             py::gil_scoped_release rel;
             c.all_gather((const void*)send, (void*)recv, bytes_per_rank,
                          as_stream(stream));
           },
           py::arg("send"), py::arg("recv"), py::arg("bytes_per_rank"),
           py::arg("stream") = 0)
      .def("reduce_scatter",
           [](Communicator& c, uintptr_t send, uintptr_t recv,
              size_t count_per_rank, int dtype, int op, uintptr_t stream) {
             py::gil_scoped_release rel;
             c.reduce_scatter((const void*)send, (void*)recv, count_per_rank,
                              dtype, op, as_stream(stream));
           },
           py::arg("send"), py::arg("recv"), py::arg("count_per_rank"),
           py::arg("dtype") = 3, py::arg("op") = 0, py::arg("stream") = 0)
      .def("send",
           [](Communicator& c, uintptr_t ptr, size_t bytes, int peer,
              uintptr_t stream) {
             py::gil_scoped_release rel;
             c.send((const void*)ptr, bytes, peer, as_stream(stream));
           },
           py::arg("ptr"), py::arg("bytes"), py::arg("peer"),
           py::arg("stream") = 0)
      .def("recv",
           [](Communicator& c, uintptr_t ptr, size_t bytes, int peer,
              uintptr_t stream) {
             py::gil_scoped_release rel;
             c.recv((void*)ptr, bytes, peer, as_stream(stream));
           },
           py::arg("ptr"), py::arg("bytes"), py::arg("peer"),
           py::arg("stream") = 0)
      .def("barrier",
           [](Communicator& c) {
             py::gil_scoped_release rel;
             c.barrier();
           })
      .def("all_reduce_scalar",
           [](Communicator& c, double v, int op) {
             py::gil_scoped_release rel;
             return c.all_reduce_scalar(v, op);
           },
           py::arg("v"), py::arg("op") = 0)
      .def("stream_synchronize", [](Communicator& c) {
        py::gil_scoped_release rel;
        c.stream_synchronize();
      });

  // ------------------------------------------------------------- engine
  py::class_<Engine, std::shared_ptr<Engine>>(m, "Engine")
      .def(py::init([](int device, py::buffer weights, size_t arena_bytes,
                       std::vector<py::dict> ops_in,
                       std::vector<std::pair<int64_t, size_t>> inputs,
                       std::vector<std::pair<int64_t, size_t>> outputs,
                       bool managed_weights) {
             py::buffer_info wi = weights.request();
             std::vector<OpDesc> ops;
             ops.reserve(ops_in.size());
             for (auto& d : ops_in) ops.push_back(op_from_dict(d));
             auto to_bindings =
                 [](const std::vector<std::pair<int64_t, size_t>>& v) {
                   std::vector<BindingDesc> out;
                   for (auto& p : v) out.push_back({p.first, p.second});
                   return out;
                 };
             return std::make_shared<Engine>(
                 device, wi.ptr, (size_t)(wi.size * wi.itemsize), arena_bytes,
                 std::move(ops), to_bindings(inputs), to_bindings(outputs),
                 managed_weights);
           }),
           py::arg("device"), py::arg("weights"), py::arg("arena_bytes"),
           py::arg("ops"), py::arg("inputs"), py::arg("outputs"),
           py::arg("managed_weights") = false)
      .def_property_readonly("device", &Engine::device)
      .def_property_readonly("arena_bytes", &Engine::arena_bytes)
      .def_property_readonly("input_bytes", &Engine::input_bytes)
      .def_property_readonly("output_bytes", &Engine::output_bytes)
      .def_property_readonly("n_inputs",
                             [](Engine& e) { return e.inputs().size(); })
      .def_property_readonly("n_outputs",
                             [](Engine& e) { return e.outputs().size(); })
      .def_property_readonly("weights_ptr", &Engine::weights_ptr)
      .def_property_readonly("weight_bytes", &Engine::weight_bytes)
      .def("upload_weights", [](Engine& e, py::buffer b) {
        py::buffer_info bi = b.request();
        e.upload_weights(bi.ptr, (size_t)(bi.size * bi.itemsize));
      });

  py::class_<ExecutionContext>(m, "ExecutionContext")
      .def(py::init<std::shared_ptr<Engine>, uintptr_t>(), py::arg("engine"),
           py::arg("external_arena") = 0)
      .def("capture",
           [](ExecutionContext& c) {
             py::gil_scoped_release rel;
             c.capture();
           })
      .def("launch", &ExecutionContext::launch)
      .def("synchronize",
           [](ExecutionContext& c) {
             py::gil_scoped_release rel;
             c.synchronize();
           })
      .def("ready", &ExecutionContext::ready)
      .def("set_timing", &ExecutionContext::set_timing)
      .def("stage_times_ms",
           [](ExecutionContext& c) {
             auto t = c.stage_times_ms();
             return py::make_tuple(t[0], t[1], t[2]);
           })
      .def_property_readonly("host_input_ptr",
                             [](ExecutionContext& c) {
                               return c.host_input_ptr(0);
                             })
      .def_property_readonly("host_output_ptr",
                             [](ExecutionContext& c) {
                               return c.host_output_ptr(0);
                             })
      .def_property_readonly("arena_ptr", &ExecutionContext::arena_ptr)
      .def("input_view",
           [](ExecutionContext& c, size_t bytes, int i) {
             return py::memoryview::from_memory((void*)c.host_input_ptr(i),
                                                bytes);
           },
           py::arg("bytes"), py::arg("i") = 0)
      .def("output_view",
           [](ExecutionContext& c, size_t bytes, int i) {
             return py::memoryview::from_memory((void*)c.host_output_ptr(i),
                                                bytes);
           },
           py::arg("bytes"), py::arg("i") = 0);
}
