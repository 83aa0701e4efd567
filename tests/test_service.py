"""Inference service tests: CPU path uses a fake manager (the reference's
echo-backend pattern); the GPU test runs the real engine behind gRPC."""
import threading
from concurrent.futures import Future

import numpy as np
import pytest

from trtlab_amd.rpc import InferRequest, InferResponse, SyncClient
from trtlab_amd.rpc.server import Server
from trtlab_amd.rpc.service import InferenceResources, InferenceService


class _FakePlan:
    input_shape = (2, 4)
    output_shape = (2, 4)
    inputs = [dict(name="input", shape=(2, 4), dtype="f16")]
    outputs = [dict(name="output", shape=(2, 4), dtype="f16")]


class _FakeEngine:
    plan = _FakePlan()


class _FakeRunner:
    """Doubles the input asynchronously (stands in for InferRunner)."""

    def infer(self, batch):
        fut = Future()

        def run():
            fut.set_result((batch.astype(np.float32) * 2).astype(np.float16))

        threading.Thread(target=run, daemon=True).start()
        return fut


class _FakeManager:
    def infer_runner(self, name):
        return _FakeRunner()

    def get_model(self, name):
        return _FakeEngine()


@pytest.fixture()
def service_server():
    server = Server("127.0.0.1:0")
    svc = InferenceService(InferenceResources(_FakeManager()))
    server.register_service(svc)
    server.register_service(svc.health_service)
    server.async_start()
    yield server
    server.shutdown()


def test_inference_service_roundtrip(service_server):
    c = SyncClient(f"127.0.0.1:{service_server.port}")
    x = np.arange(8, dtype=np.float16).reshape(2, 4)
    resp = c.call("trtlab.Inference", "Compute",
                  InferRequest(model="m", input=x.tobytes(), shape=[2, 4],
                               dtype="f16", batch_id=5),
                  InferResponse, timeout=10)
    out = np.frombuffer(resp.output, dtype=np.float16).reshape(2, 4)
    assert np.allclose(out, x * 2)
    assert resp.batch_id == 5
    assert resp.compute_ms >= 0
    assert resp.request_ms >= resp.compute_ms
    c.close()


def test_metrics_observe():
    from trtlab_amd.utils.metrics import Metrics

    m = Metrics(port=0)  # not started: observe() only
    m.observe(2.0, 3.0)
    m.observe(1.0, 50.0)  # queueing-dominated: load ratio 50 -> last bucket


@pytest.mark.gpu
def test_inference_service_gpu_end_to_end():
    from trtlab_amd.engine.planner import Planner
    from trtlab_amd.engine.reference import run_reference
    from trtlab_amd.engine.runtime import InferenceManager
    from trtlab_amd.models import build_resnet

    g = build_resnet(50, batch=2, image=64, seed=0)
    plan = Planner().compile(g)
    mgr = InferenceManager(max_contexts=2)
    mgr.register_model("rn50", plan)
    mgr.allocate_resources()

    server = Server("127.0.0.1:0")
    svc = InferenceService(InferenceResources(mgr))
    server.register_service(svc)
    server.async_start()
    try:
        c = SyncClient(f"127.0.0.1:{server.port}")
        x = (np.random.RandomState(3).randn(*plan.input_shape) * 0.5).astype(np.float16)
        resp = c.call("trtlab.Inference", "Compute",
                      InferRequest(model="rn50", input=x.tobytes(),
                                   shape=list(plan.input_shape), dtype="f16"),
                      InferResponse, timeout=120)
        out = np.frombuffer(resp.output, dtype=np.float16).reshape(
            tuple(resp.shape)).astype(np.float32)
        ref = run_reference(plan, x.astype(np.float32))
        err = np.abs(out - ref).max() / max(np.abs(ref).max(), 1e-6)
        assert err < 0.08, err
        c.close()
    finally:
        server.shutdown()
        mgr.shutdown()


def test_metrics_multiple_instances():
    from trtlab_amd.utils.metrics import Metrics

    m1 = Metrics(port=0)
    m2 = Metrics(port=0)  # own registries: no duplicate-collector error
    m1.observe(1.0, 2.0)
    m2.observe(1.0, 2.0)


@pytest.mark.gpu
def test_multi_model_manager():
    """Two models behind one InferenceManager (reference RegisterModel xN)."""
    from trtlab_amd.engine.planner import Planner
    from trtlab_amd.engine.runtime import InferenceManager
    from trtlab_amd.models import build_resnet

    g1 = build_resnet(50, batch=1, image=64, seed=0)
    g2 = build_resnet(50, batch=2, image=64, seed=1)
    mgr = InferenceManager(max_contexts=1)
    mgr.register_model("a", Planner().compile(g1))
    mgr.register_model("b", Planner().compile(g2))
    mgr.allocate_resources()
    ra, rb = mgr.infer_runner("a"), mgr.infer_runner("b")
    xa = np.random.RandomState(0).randn(1, 64, 64, 3).astype(np.float32)
    xb = np.random.RandomState(1).randn(2, 64, 64, 3).astype(np.float32)
    oa = ra.infer(xa).result(60)
    ob = rb.infer(xb).result(60)
    assert oa.shape == (1, 1000) and ob.shape == (2, 1000)
    mgr.shutdown()


def test_shm_zero_copy_input(service_server):
    """POSIX shared-memory input path (reference SharedMemoryService)."""
    from trtlab_amd.rpc import ShmInput

    c = SyncClient(f"127.0.0.1:{service_server.port}")
    x = np.arange(8, dtype=np.float16).reshape(2, 4)
    with ShmInput(x) as shm:
        resp = c.call("trtlab.Inference", "Compute",
                      InferRequest(model="m", shape=[2, 4], dtype="f16",
                                   shm_name=shm.name, shm_size=shm.size),
                      InferResponse, timeout=10)
    out = np.frombuffer(resp.output, dtype=np.float16).reshape(2, 4)
    assert np.allclose(out, x * 2)
    c.close()


def test_remote_inference_manager(service_server):
    """Client-side manager mirroring the local surface (reference
    PyRemoteInferenceManager)."""
    from trtlab_amd.rpc.remote import RemoteInferenceManager

    m = RemoteInferenceManager(f"127.0.0.1:{service_server.port}")
    assert m.ready()
    runner = m.infer_runner("m")
    x = np.arange(8, dtype=np.float16).reshape(2, 4)
    out = runner.infer(x).result(timeout=10)
    assert np.allclose(out, x * 2)
    # shm transport variant
    out2 = m.infer_runner("m", use_shm=True).infer(x).result(timeout=10)
    assert np.allclose(out2, x * 2)
    m.close()


def test_unknown_model_is_an_rpc_error():
    """Requests for unregistered models surface as client-side errors, not
    server crashes; the service keeps serving afterwards."""
    class _StrictManager(_FakeManager):
        def infer_runner(self, name):
            if name != "m":
                raise KeyError(name)
            return _FakeRunner()

        def get_model(self, name):
            if name != "m":
                raise KeyError(name)
            return _FakeEngine()

    server = Server("127.0.0.1:0")
    svc = InferenceService(InferenceResources(_StrictManager()))
    server.register_service(svc)
    server.register_service(svc.health_service)
    server.async_start()
    try:
        c = SyncClient(f"127.0.0.1:{server.port}")
        bad = InferRequest(model="nope", shape=[2, 4], dtype="f16",
                           input=np.zeros(8, np.float16).tobytes())
        with pytest.raises(Exception):
            c.call("trtlab.Inference", "Compute", bad, InferResponse,
                   timeout=5)
        # still alive after the error
        x = np.arange(8, dtype=np.float16).reshape(2, 4)
        good = InferRequest(model="m", input=x.tobytes(), shape=[2, 4],
                            dtype="f16")
        r = c.call("trtlab.Inference", "Compute", good, InferResponse,
                   timeout=10)
        assert list(r.shape) == [2, 4]
    finally:
        server.shutdown()


class _FakeMultiPlan:
    input_shape = (2, 4)
    output_shape = (2, 4)
    inputs = [dict(name="a", shape=(2, 4), dtype="f16"),
              dict(name="b", shape=(2, 4), dtype="f16")]
    outputs = [dict(name="sum", shape=(2, 4), dtype="f16"),
               dict(name="diff", shape=(2, 4), dtype="f16")]


class _FakeMultiEngine:
    plan = _FakeMultiPlan()


class _FakeMultiRunner:
    """sum/diff of two named inputs (multi-binding InferRunner stand-in)."""

    def infer(self, batch):
        fut = Future()
        a, b = batch["a"].astype(np.float32), batch["b"].astype(np.float32)
        fut.set_result({"sum": (a + b).astype(np.float16),
                        "diff": (a - b).astype(np.float16)})
        return fut


class _FakeMultiManager:
    def infer_runner(self, name):
        return _FakeMultiRunner()

    def get_model(self, name):
        return _FakeMultiEngine()


def test_named_tensor_rpc_roundtrip():
    """Named multi-binding tensors ride the InferRequest.inputs /
    InferResponse.outputs fields (the gRPC service carries
    named tensors)."""
    from trtlab_amd.rpc import NamedTensor

    server = Server("127.0.0.1:0")
    svc = InferenceService(InferenceResources(_FakeMultiManager()))
    server.register_service(svc)
    server.async_start()
    try:
        c = SyncClient(f"127.0.0.1:{server.port}")
        rng = np.random.RandomState(0)
        a = rng.randn(2, 4).astype(np.float16)
        b = rng.randn(2, 4).astype(np.float16)
        req = InferRequest(
            model="m", batch_id=9,
            inputs=[NamedTensor(name="a", data=a.tobytes(), shape=[2, 4],
                                dtype="f16"),
                    NamedTensor(name="b", data=b.tobytes(), shape=[2, 4],
                                dtype="f16")])
        resp = c.call("trtlab.Inference", "Compute", req, InferResponse)
        outs = {t.name: np.frombuffer(t.data, np.float16).reshape(
            tuple(t.shape)) for t in resp.outputs}
        assert set(outs) == {"sum", "diff"}
        np.testing.assert_allclose(outs["sum"],
                                   (a.astype(np.float32) +
                                    b.astype(np.float32)).astype(np.float16))
        np.testing.assert_allclose(outs["diff"],
                                   (a.astype(np.float32) -
                                    b.astype(np.float32)).astype(np.float16))
        # primary output mirrors outputs[0] for single-output clients
        prim = np.frombuffer(resp.output, np.float16).reshape(2, 4)
        np.testing.assert_allclose(prim, outs["sum"])
    finally:
        server.shutdown()


def test_trtis_surface_roundtrip():
    """The TRTIS v1 GRPCService (nvidia.inferenceserver package, exact
    reference field numbers): Status, Health and Infer with raw_input /
    InferRequestHeader all answer correctly over a real loopback server."""
    from trtlab_amd.rpc.trtis import (InferRequestHeader, StatusRequest,
                                      StatusResponse, TrtisHealthRequest,
                                      TrtisHealthResponse, TrtisInferRequest,
                                      TrtisInferResponse, TrtisService)

    server = Server("127.0.0.1:0")
    svc = TrtisService(InferenceResources(_FakeManager()))
    server.register_service(svc.service)
    server.async_start()
    try:
        c = SyncClient(f"127.0.0.1:{server.port}")
        st = c.call("nvidia.inferenceserver.GRPCService", "Status",
                    StatusRequest(), StatusResponse)
        assert st.request_status.code == 1  # SUCCESS
        assert st.server_status.ready_state == 2  # SERVER_READY
        h = c.call("nvidia.inferenceserver.GRPCService", "Health",
                   TrtisHealthRequest(mode="ready"), TrtisHealthResponse)
        assert h.health
        x = np.arange(8, dtype=np.float16).reshape(2, 4)
        hdr = InferRequestHeader(batch_size=2)
        i = hdr.input.add()
        i.name = "input"
        i.byte_size = x.nbytes
        req = TrtisInferRequest(model_name="m", meta_data=hdr,
                                raw_input=[x.tobytes()], batch_id=7)
        resp = c.call("nvidia.inferenceserver.GRPCService", "Infer", req,
                      TrtisInferResponse)
        assert resp.request_status.code == 1
        assert resp.batch_id == 7
        out = np.frombuffer(resp.raw_output[0], np.float16).reshape(2, 4)
        np.testing.assert_allclose(out, x * 2)
        assert resp.meta_data.output[0].raw.byte_size == out.nbytes
    finally:
        server.shutdown()


def test_trtis_model_config_generator():
    """model_config.pbtxt generation from a compiled plan (reference
    12_ConfigGenerator role): bindings, dtype mapping, batch-dim
    stripping, instance group and dynamic batching stanzas."""
    from trtlab_amd.engine.planner import Planner
    from trtlab_amd.models import build_bert
    from trtlab_amd.rpc.trtis import model_config_pbtxt

    g = build_bert(batch=2, seq=64, layers=1, seed=0,
                   embeddings=True, varlen=True, mask_input=True)
    plan = Planner().compile(g)
    txt = model_config_pbtxt(plan, "bert", max_batch_size=16, instances=3,
                             preferred_batch_sizes=(4, 8),
                             queue_delay_us=200)
    assert 'name: "bert"' in txt
    assert "max_batch_size: 16" in txt
    assert txt.count("input {") == len(plan.inputs)
    assert txt.count("output {") == len(plan.outputs)
    assert "TYPE_INT32" in txt          # token-id inputs
    assert "TYPE_FP16" in txt           # hidden outputs
    assert "count: 3" in txt and "KIND_GPU" in txt
    assert "preferred_batch_size: 4" in txt
    assert "max_queue_delay_microseconds: 200" in txt
    # max_batch strips the leading batch dim from a [M, H] binding
    out_b = plan.outputs[0]
    dims = [str(d) for d in out_b["shape"][1:]]
    assert f"dims: [ {', '.join(dims)} ]" in txt
