"""GPU runtime wrapper: EnginePlan -> native Engine + pooled, graph-captured
ExecutionContexts + the async 3-stage infer pipeline.

Reference mapping:
  NativeEngine        ~ Runtime::DeserializeEngine + Model (runtime.h:43)
  NativeContext       ~ StaticSingleModelGraphWorkspace (workspace.cc:21-75)
  InferenceManager    ~ v1 InferenceManager (inference_manager.cc:254-312)
  InferRunner         ~ infer_runner.h:37 (pre -> hip -> post pipeline)
  InferBench          ~ infer_bench.h:48
"""
from __future__ import annotations

import time
import threading
from concurrent.futures import Future
from typing import Dict, List, Optional

import numpy as np

from trtlab_amd import native
from trtlab_amd.core import Pool, ThreadPool
from trtlab_amd.engine.planner import EnginePlan


class NativeEngine:
    """Compiled model resident on one GPU (weights + op plan)."""

    def __init__(self, plan: EnginePlan, device: int = 0,
                 autotune: bool = False, managed_weights: bool = False):
        """managed_weights: weight blob in hipMallocManaged memory advised
        read-mostly (reference ManagedRuntime / NvAllocator weights path,
        trtlab/tensorrt/src/allocator.cc:12-56). Default off — explicit
        HBM residency is faster and MI355X has 288 GB per GPU."""
        self._C = native()
        if self._C.hip.device_count() == 0:
            raise RuntimeError(
                "NativeEngine requires a GPU; the HIP extension found no "
                "devices (do NOT fall back to eager torch on a GPU box)")
        if autotune:
            # builder-time tactic selection (TensorRT-builder role); the
            # chosen tile codes are baked into plan.ops and persist through
            # the plan cache
            from trtlab_amd.engine.autotune import autotune_plan

            autotune_plan(plan, device=device)
        self.plan = plan
        self.device = device
        self.engine = self._C.Engine(
            device, plan.weights, plan.arena_bytes, plan.ops,
            [(b["off"], b["bytes"]) for b in plan.inputs],
            [(b["off"], b["bytes"]) for b in plan.outputs],
            managed_weights=managed_weights)

    def upload_weights(self, blob: np.ndarray) -> None:
        self.engine.upload_weights(blob)

    def create_context(self, capture: bool = True, timed: bool = False,
                       arena_ptr: int = 0) -> "NativeContext":
        """arena_ptr: optional caller-owned device memory for the
        activation arena (carve from a shared memory.DeviceArena so
        multiple models serve from one pool — reference growing
        block_arena / bfit role)."""
        return NativeContext(self, capture=capture, timed=timed,
                             arena_ptr=arena_ptr)


class NativeContext:
    """Private stream + activation arena + pinned bindings + hipGraph."""

    def __init__(self, engine: NativeEngine, capture: bool = True,
                 timed: bool = False, arena_ptr: int = 0):
        self._C = native()
        self.engine = engine
        self.plan = engine.plan
        self.ctx = self._C.ExecutionContext(engine.engine, arena_ptr)
        if timed:
            # per-stage H2D/compute/D2H events (reference
            # TimedBenchmarkWorkspace); recorded inside the captured graph
            self.ctx.set_timing(True)
        # bf16 plans carry bf16 bit patterns in the bindings (numpy has no
        # bf16 dtype); infer() converts via torch at the edges
        np_dt = {"f16": np.float16, "bf16": np.int16, "i32": np.int32,
                 "f32": np.float32, "i8": np.int8}

        # N named bindings, pinned zero-copy views (reference Bindings
        # carving, bindings.h:60-120); index 0 = primary binding.
        self.inputs: Dict[str, np.ndarray] = {}
        self._in_bf16_map: Dict[str, bool] = {}
        for i, b in enumerate(self.plan.inputs):
            v = np.frombuffer(self.ctx.input_view(b["bytes"], i),
                              dtype=np_dt[b["dtype"]]).reshape(b["shape"])
            self.inputs[b["name"]] = v
            self._in_bf16_map[b["name"]] = b["dtype"] == "bf16"
        self.outputs: Dict[str, np.ndarray] = {}
        for i, b in enumerate(self.plan.outputs):
            v = np.frombuffer(self.ctx.output_view(b["bytes"], i),
                              dtype=np_dt[b["dtype"]]).reshape(b["shape"])
            self.outputs[b["name"]] = v

        in0 = self.plan.inputs[0]["name"]
        self._in_bf16 = self._in_bf16_map[in0]
        self._out_bf16 = self.plan.outputs[0]["dtype"] == "bf16"
        self._in_view = self.inputs[in0]
        self._out_view = self.outputs[self.plan.outputs[0]["name"]]
        if capture:
            self.ctx.capture()

    @property
    def input(self) -> np.ndarray:
        """Pinned host input binding (write your batch here)."""
        return self._in_view

    @property
    def output(self) -> np.ndarray:
        """Pinned host output binding (valid after synchronize)."""
        return self._out_view

    def write_input(self, batch, name: Optional[str] = None) -> None:
        """Copy a host batch into a pinned input binding (primary binding
        by default; pass `name` or a {name: array} dict for multi-input
        models). Handles the bf16 bit-pattern representation (numpy has no
        bf16 dtype — float inputs are converted to bf16 BITS, never
        numerically cast to int16) and validates shape up front so
        malformed requests fail loudly."""
        if isinstance(batch, dict):
            for k, v in batch.items():
                self.write_input(v, name=k)
            return
        view = self._in_view if name is None else self.inputs[name]
        bf16 = self._in_bf16 if name is None else self._in_bf16_map[name]
        if tuple(batch.shape) != tuple(view.shape):
            raise ValueError(
                f"input shape {tuple(batch.shape)} != plan shape "
                f"{tuple(view.shape)} for binding "
                f"{name or self.plan.inputs[0]['name']}")
        if bf16 and batch.dtype != np.int16:
            from trtlab_amd.engine.planner import _bf16_bits

            np.copyto(view, _bf16_bits(batch).reshape(view.shape))
        else:
            np.copyto(view, batch.astype(view.dtype, copy=False))

    def infer(self, batch=None):
        """Synchronous convenience path; returns a COPY of the primary
        output (the zero-copy `.output` view is only valid while this
        context lives). Multi-output models: use infer_all()."""
        if batch is not None:
            self.write_input(batch)
        self.ctx.launch()
        self.ctx.synchronize()
        out = np.array(self._out_view, copy=True)
        if self._out_bf16:
            import torch

            out = torch.from_numpy(out).view(torch.bfloat16).to(
                torch.float32).numpy()
        return out

    def infer_all(self, batch=None) -> Dict[str, np.ndarray]:
        """Run one forward and return copies of ALL output bindings by
        name (3-input/2-output models etc.)."""
        if batch is not None:
            self.write_input(batch)
        self.ctx.launch()
        self.ctx.synchronize()
        return {k: np.array(v, copy=True) for k, v in self.outputs.items()}

    def launch(self):
        self.ctx.launch()

    def synchronize(self):
        self.ctx.synchronize()

    def ready(self) -> bool:
        return self.ctx.ready()

    def stage_times_ms(self) -> tuple:
        """(h2d_ms, compute_ms, d2h_ms) of the last completed launch
        (requires timed=True)."""
        return self.ctx.stage_times_ms()


class InferenceManager:
    """Resource orchestrator: per-model context pools + named thread pools
    (reference inference_manager.cc:254-312: two-level concurrency limiter +
    'pre'/'cuda'/'post' pools — ours are 'pre'/'hip'/'post')."""

    def __init__(self, max_contexts: int = 2, device: int = 0,
                 pre_threads: int = 1, hip_threads: int = 1,
                 post_threads: int = 2, shared_arena: bool = False,
                 arena_max_bytes: int = 0, max_executions: int = 0):
        """shared_arena=True: all models' activation arenas are carved from
        ONE growing best-fit device pool (native DeviceArena — reference
        bfit_allocator/block_arena role) instead of per-context hipMallocs;
        pool stats (high-water, histogram) export via arena_stats().

        max_executions: GLOBAL cap on concurrently-launched forwards across
        ALL models — the reference's two-level concurrency limiter
        (inference_manager.cc:254-282: the global ExecutionContext pool
        gates on top of the per-model context pools). 0 = per-model pools
        only."""
        import threading as _threading

        self.device = device
        self.max_contexts = max_contexts
        self._exec_sem = (_threading.Semaphore(max_executions)
                          if max_executions > 0 else None)
        self._models: Dict[str, NativeEngine] = {}
        self._ctx_pools: Dict[str, Pool] = {}
        self.arena = None
        self._arena_allocs: List[int] = []
        if shared_arena:
            from trtlab_amd import native

            self.arena = native().memory.DeviceArena(
                device, 0, arena_max_bytes)
        self.thread_pools: Dict[str, ThreadPool] = {
            "pre": ThreadPool(pre_threads, "pre"),
            "hip": ThreadPool(hip_threads, "hip"),
            "post": ThreadPool(post_threads, "post"),
        }

    def register_model(self, name: str, plan: EnginePlan) -> None:
        self._models[name] = NativeEngine(plan, self.device)

    def allocate_resources(self) -> None:
        for name, eng in self._models.items():
            if name not in self._ctx_pools:
                ptrs = [0] * self.max_contexts
                if self.arena is not None:
                    ptrs = [self.arena.allocate(eng.plan.arena_bytes)
                            for _ in range(self.max_contexts)]
                    self._arena_allocs.extend(ptrs)
                self._ctx_pools[name] = Pool(
                    [eng.create_context(arena_ptr=p) for p in ptrs])

    def arena_stats(self) -> Optional[dict]:
        return self.arena.stats() if self.arena is not None else None

    def get_model(self, name: str) -> NativeEngine:
        return self._models[name]

    def infer_runner(self, name: str) -> "InferRunner":
        return InferRunner(self, name)

    # ---- the PyInferenceManager surface (reference pybind/trtlab/infer.cc:
    # register_tensorrt_engine / update_resources / serve) ----
    def register_onnx(self, name: str, path: str, batch: int = 8,
                      dtype: int = 0) -> None:
        from trtlab_amd.engine.onnx_io import load_onnx
        from trtlab_amd.engine.planner import Planner

        self.register_model(name, Planner(dtype=dtype).compile(
            load_onnx(path, batch=batch)))

    def register_plan_file(self, name: str, path: str) -> None:
        from trtlab_amd.engine.plan_io import load_plan

        self.register_model(name, load_plan(path))

    def serve(self, port: int = 50051, metrics_port: int = 0):
        """Start a gRPC inference service over the registered models;
        returns the running Server (reference PyInferenceManager::Serve)."""
        from trtlab_amd.rpc.server import Server
        from trtlab_amd.rpc.service import InferenceResources, InferenceService
        from trtlab_amd.utils.metrics import Metrics

        metrics = Metrics.initialize(metrics_port) if metrics_port else None
        resources = InferenceResources(self)
        svc = InferenceService(resources, metrics=metrics)
        server = Server(f"0.0.0.0:{port}")
        server.register_service(svc)
        server.register_service(svc.health_service)
        # TRTIS v1 surface (nvidia.inferenceserver.GRPCService) so stock
        # TRTIS clients interoperate (reference 11_Protos API)
        from trtlab_amd.rpc.trtis import TrtisService

        server.register_service(TrtisService(resources).service)
        server.async_start()
        return server

    def shutdown(self):
        for p in self.thread_pools.values():
            p.shutdown()


class InferRunner:
    """Async 3-stage pipeline: pre(copy into pinned bindings) -> hip(launch
    graph) -> post(sync, hand result to completion) — returns a Future
    (reference infer_runner.h:75-101 Enqueue chain)."""

    def __init__(self, manager: InferenceManager, model: str):
        self.manager = manager
        self.model = model
        self._pool = manager._ctx_pools[model]

    def infer(self, batch: np.ndarray) -> Future:
        fut: Future = Future()
        tp = self.manager.thread_pools

        # Every stage propagates failures into `fut` and releases the
        # checkout — a malformed request (bad shape/dtype) must reject its
        # own Future, never hang the awaiting RPC handler.
        def pre():
            co = self._pool.pop()  # blocks: concurrency limiter
            try:
                ctx: NativeContext = co.item
                ctx.write_input(batch)  # bf16-bit aware + shape-validated
                tp["hip"].enqueue(hip_stage, co)
            except BaseException as e:  # noqa: BLE001
                fut.set_exception(e)
                co.release()

        sem = self.manager._exec_sem

        def hip_stage(co):
            try:
                if sem is not None:
                    sem.acquire()  # level 2: global cross-model cap
                co.item.launch()
                tp["post"].enqueue(post, co)
            except BaseException as e:  # noqa: BLE001
                if sem is not None:
                    sem.release()
                fut.set_exception(e)
                co.release()

        def post(co):
            try:
                out = None
                try:
                    co.item.synchronize()
                    if len(co.item.plan.outputs) > 1:
                        out = {k: np.array(v, copy=True)
                               for k, v in co.item.outputs.items()}
                    else:
                        out = np.array(co.item.output, copy=True)
                finally:
                    if sem is not None:
                        sem.release()
                    co.release()
                fut.set_result(out)
            except BaseException as e:  # noqa: BLE001
                fut.set_exception(e)

        tp["pre"].enqueue(pre)
        return fut


class InferBench:
    """Throughput/latency harness: saturate the context pool for N seconds,
    report inf/sec + latency quantiles (reference infer_bench.h:48)."""

    def __init__(self, manager: InferenceManager, model: str):
        self.runner = manager.infer_runner(model)
        self.batch_size = manager.get_model(model).plan.input_shape[0]

    def run(self, batch: np.ndarray, seconds: float = 5.0,
            max_outstanding: int = 8) -> Dict[str, float]:
        lat: List[float] = []
        lock = threading.Lock()
        inflight = threading.Semaphore(max_outstanding)
        start = time.monotonic()
        count = 0
        futures = []
        while time.monotonic() - start < seconds:
            inflight.acquire()
            t0 = time.monotonic()

            def done(f, t0=t0):
                with lock:
                    lat.append(time.monotonic() - t0)
                inflight.release()

            f = self.runner.infer(batch)
            f.add_done_callback(done)
            futures.append(f)
            count += 1
        for f in futures:
            f.result()
        elapsed = time.monotonic() - start
        lat_ms = np.array(sorted(lat)) * 1e3
        return dict(
            batches=count,
            seconds=elapsed,
            batches_per_sec=count / elapsed,
            inf_per_sec=count * self.batch_size / elapsed,
            p50_ms=float(np.percentile(lat_ms, 50)),
            p90_ms=float(np.percentile(lat_ms, 90)),
            p99_ms=float(np.percentile(lat_ms, 99)),
        )
