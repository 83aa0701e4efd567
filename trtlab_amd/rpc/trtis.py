"""TRTIS-compatible inference service surface.

Wire-compatible with the reference's `nvidia.inferenceserver.GRPCService`
v1 API (examples/11_Protos/inference/nvidia_inference.proto + api.proto +
request_status.proto — schema SHAPE mirrored for interop, implementation
original): standard TRTIS clients (the reference's own pybind
RemoteInferenceManager, 02_TensorRT_GRPC clients) can Status/Health/Infer
against this server unchanged. Built with dynamic descriptors (no protoc
in the offline image), same technique as rpc/proto.py.

Field numbers match the reference protos exactly:
  InferRequest   { model_name 1, version 2, meta_data 3, raw_input 4,
                   batch_id 100, batch_size 101 }
  InferResponse  { request_status 1, meta_data 2, raw_output 3,
                   batch_id 100, compute_time 101, request_time 102 }
  InferRequestHeader  { batch_size 1, input 2 {name 1, byte_size 2},
                        output 3 {name 1, byte_size 2} }
  InferResponseHeader { model_name 1, model_version 2, batch_size 3,
                        output 4 {name 1, raw 2 {byte_size 1}} }
  RequestStatus  { code 1 (SUCCESS=1), msg 2, server_id 3, request_id 4 }
"""
from __future__ import annotations

import asyncio
import time
from typing import Dict

import numpy as np
from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_pool = descriptor_pool.Default()
_T = descriptor_pb2.FieldDescriptorProto

_f = descriptor_pb2.FileDescriptorProto()
_f.name = "trtlab_amd/rpc/nvidia_inference.proto"
_f.package = "nvidia.inferenceserver"
_f.syntax = "proto3"


def _field(m, name, num, ftype, repeated=False, type_name=None):
    fd = m.field.add()
    fd.name = name
    fd.number = num
    fd.type = ftype
    if type_name:
        fd.type_name = type_name
    fd.label = _T.LABEL_REPEATED if repeated else _T.LABEL_OPTIONAL


def _msg(name):
    m = _f.message_type.add()
    m.name = name
    return m


_rs = _msg("RequestStatus")
_field(_rs, "code", 1, _T.TYPE_INT32)  # enum on the wire = varint
_field(_rs, "msg", 2, _T.TYPE_STRING)
_field(_rs, "server_id", 3, _T.TYPE_STRING)
_field(_rs, "request_id", 4, _T.TYPE_UINT64)

_ss = _msg("ServerStatus")
_field(_ss, "id", 1, _T.TYPE_STRING)
_field(_ss, "version", 2, _T.TYPE_STRING)
_field(_ss, "uptime_ns", 3, _T.TYPE_UINT64)
_field(_ss, "ready_state", 7, _T.TYPE_INT32)  # SERVER_READY = 2

_sr = _msg("StatusRequest")
_field(_sr, "model_name", 1, _T.TYPE_STRING)

_sresp = _msg("StatusResponse")
_field(_sresp, "request_status", 1, _T.TYPE_MESSAGE,
       type_name=".nvidia.inferenceserver.RequestStatus")
_field(_sresp, "server_status", 2, _T.TYPE_MESSAGE,
       type_name=".nvidia.inferenceserver.ServerStatus")

_hr = _msg("HealthRequest")
_field(_hr, "mode", 1, _T.TYPE_STRING)

_hresp = _msg("HealthResponse")
_field(_hresp, "request_status", 1, _T.TYPE_MESSAGE,
       type_name=".nvidia.inferenceserver.RequestStatus")
_field(_hresp, "health", 2, _T.TYPE_BOOL)

_irh = _msg("InferRequestHeader")
_in = _irh.nested_type.add()
_in.name = "Input"
_field(_in, "name", 1, _T.TYPE_STRING)
_field(_in, "byte_size", 2, _T.TYPE_UINT64)
_out = _irh.nested_type.add()
_out.name = "Output"
_field(_out, "name", 1, _T.TYPE_STRING)
_field(_out, "byte_size", 2, _T.TYPE_UINT64)
_field(_irh, "batch_size", 1, _T.TYPE_UINT32)
_field(_irh, "input", 2, _T.TYPE_MESSAGE, repeated=True,
       type_name=".nvidia.inferenceserver.InferRequestHeader.Input")
_field(_irh, "output", 3, _T.TYPE_MESSAGE, repeated=True,
       type_name=".nvidia.inferenceserver.InferRequestHeader.Output")

_irsp = _msg("InferResponseHeader")
_o2 = _irsp.nested_type.add()
_o2.name = "Output"
_raw = _o2.nested_type.add()
_raw.name = "Raw"
_field(_raw, "byte_size", 1, _T.TYPE_UINT64)
_field(_o2, "name", 1, _T.TYPE_STRING)
_field(_o2, "raw", 2, _T.TYPE_MESSAGE,
       type_name=".nvidia.inferenceserver.InferResponseHeader.Output.Raw")
_field(_irsp, "model_name", 1, _T.TYPE_STRING)
_field(_irsp, "model_version", 2, _T.TYPE_UINT32)
_field(_irsp, "batch_size", 3, _T.TYPE_UINT32)
_field(_irsp, "output", 4, _T.TYPE_MESSAGE, repeated=True,
       type_name=".nvidia.inferenceserver.InferResponseHeader.Output")

_ireq = _msg("InferRequest")
_field(_ireq, "model_name", 1, _T.TYPE_STRING)
_field(_ireq, "version", 2, _T.TYPE_STRING)
_field(_ireq, "meta_data", 3, _T.TYPE_MESSAGE,
       type_name=".nvidia.inferenceserver.InferRequestHeader")
_field(_ireq, "raw_input", 4, _T.TYPE_BYTES, repeated=True)
_field(_ireq, "batch_id", 100, _T.TYPE_UINT64)
_field(_ireq, "batch_size", 101, _T.TYPE_UINT32)

_iresp = _msg("InferResponse")
_field(_iresp, "request_status", 1, _T.TYPE_MESSAGE,
       type_name=".nvidia.inferenceserver.RequestStatus")
_field(_iresp, "meta_data", 2, _T.TYPE_MESSAGE,
       type_name=".nvidia.inferenceserver.InferResponseHeader")
_field(_iresp, "raw_output", 3, _T.TYPE_BYTES, repeated=True)
_field(_iresp, "batch_id", 100, _T.TYPE_UINT64)
_field(_iresp, "compute_time", 101, _T.TYPE_FLOAT)
_field(_iresp, "request_time", 102, _T.TYPE_FLOAT)

_pool.Add(_f)


def _cls(name):
    return message_factory.GetMessageClass(
        _pool.FindMessageTypeByName(f"nvidia.inferenceserver.{name}"))


RequestStatus = _cls("RequestStatus")
ServerStatus = _cls("ServerStatus")
StatusRequest = _cls("StatusRequest")
StatusResponse = _cls("StatusResponse")
TrtisHealthRequest = _cls("HealthRequest")
TrtisHealthResponse = _cls("HealthResponse")
InferRequestHeader = _cls("InferRequestHeader")
InferResponseHeader = _cls("InferResponseHeader")
TrtisInferRequest = _cls("InferRequest")
TrtisInferResponse = _cls("InferResponse")

SUCCESS = 1
SERVER_READY = 2
SERVER_ID = "trtlab_amd"


class TrtisService:
    """`nvidia.inferenceserver.GRPCService` over an InferenceResources —
    the TRTIS v1 surface the reference's example clients speak
    (pybind RemoteInferenceManager, 30_PyTensorRT client.py)."""

    def __init__(self, resources):
        from trtlab_amd.rpc.server import AsyncService

        self.resources = resources
        self._t0 = time.monotonic_ns()
        svc = AsyncService("nvidia.inferenceserver.GRPCService",
                           resources)
        svc.register_unary("Status", self._status, StatusRequest,
                           StatusResponse)
        svc.register_unary("Health", self._health, TrtisHealthRequest,
                           TrtisHealthResponse)
        svc.register_unary("Infer", self._infer, TrtisInferRequest,
                           TrtisInferResponse)
        self.service = svc

    def _ok(self, request_id: int = 0):
        return RequestStatus(code=SUCCESS, server_id=SERVER_ID,
                             request_id=request_id)

    async def _status(self, request, context, resources):
        return StatusResponse(
            request_status=self._ok(),
            server_status=ServerStatus(
                id=SERVER_ID, version="2.0",
                uptime_ns=time.monotonic_ns() - self._t0,
                ready_state=SERVER_READY))

    async def _health(self, request, context, resources):
        return TrtisHealthResponse(request_status=self._ok(), health=True)

    async def _infer(self, request, context, resources):
        t_start = time.monotonic()
        runner = resources.runner(request.model_name)
        plan = resources.manager.get_model(request.model_name).plan
        np_dt = {"f16": np.float16, "i32": np.int32, "f32": np.float32,
                 "bf16": np.int16, "i8": np.int8}
        # raw_input[i] pairs with meta_data.input[i] (TRTIS contract);
        # with no meta-data names, inputs are taken in plan binding order
        raws = list(request.raw_input)
        names = [i.name for i in request.meta_data.input]
        batch: Dict[str, np.ndarray] = {}
        for i, b in enumerate(plan.inputs):
            if i >= len(raws):
                break
            nm = names[i] if i < len(names) and names[i] else b["name"]
            bind = next((x for x in plan.inputs if x["name"] == nm), b)
            batch[bind["name"]] = np.frombuffer(
                raws[i], dtype=np_dt[bind["dtype"]]).reshape(bind["shape"])
        feed = (batch if len(plan.inputs) > 1
                else batch[plan.inputs[0]["name"]])
        fut = runner.infer(feed)
        out = await asyncio.wrap_future(fut)
        compute_s = time.monotonic() - t_start
        outs = out if isinstance(out, dict) else {
            plan.outputs[0]["name"]: out}
        hdr = InferResponseHeader(
            model_name=request.model_name, model_version=1,
            batch_size=request.meta_data.batch_size or
            int(plan.input_shape[0]))
        raw_out = []
        for b in plan.outputs:
            arr = outs[b["name"]]
            raw_out.append(arr.tobytes())
            o = hdr.output.add()
            o.name = b["name"]
            o.raw.byte_size = arr.nbytes
        return TrtisInferResponse(
            request_status=self._ok(request.batch_id),
            meta_data=hdr, raw_output=raw_out, batch_id=request.batch_id,
            compute_time=compute_s,
            request_time=time.monotonic() - t_start)


# --------------------------------------------------------------- config gen
_TRTIS_DT = {"f16": "TYPE_FP16", "bf16": "TYPE_BF16", "f32": "TYPE_FP32",
             "i32": "TYPE_INT32", "i8": "TYPE_INT8"}


def model_config_pbtxt(plan, name: str, max_batch_size: int = 0,
                       instances: int = 1,
                       preferred_batch_sizes=(), queue_delay_us: int = 0,
                       ) -> str:
    """Generate a TRTIS `model_config.pbtxt` for a compiled EnginePlan —
    the reference ships a standalone ConfigGenerator app for this
    (examples/12_ConfigGenerator; model_config.proto in
    11_Protos). Bindings come from the plan's typed input/output tables;
    dims drop the leading batch dim when max_batch_size > 0 (the TRTIS
    convention). preferred_batch_sizes/queue_delay_us emit the
    dynamic_batching stanza used with the batching server."""
    def stanza(kind, b):
        dims = list(b["shape"])
        if max_batch_size > 0 and len(dims) > 1:
            dims = dims[1:]
        dd = ", ".join(str(int(d)) for d in dims)
        return (f"{kind} {{\n  name: \"{b['name']}\"\n"
                f"  data_type: {_TRTIS_DT[b['dtype']]}\n"
                f"  dims: [ {dd} ]\n}}\n")

    out = [f"name: \"{name}\"",
           "platform: \"trtlab_amd\"",
           f"max_batch_size: {max_batch_size}", ""]
    for b in plan.inputs:
        out.append(stanza("input", b))
    for b in plan.outputs:
        out.append(stanza("output", b))
    out.append("instance_group {\n  count: %d\n  kind: KIND_GPU\n}\n"
               % instances)
    if preferred_batch_sizes:
        pb = "\n".join(f"  preferred_batch_size: {int(p)}"
                       for p in preferred_batch_sizes)
        out.append("dynamic_batching {\n%s\n  max_queue_delay_microseconds:"
                   " %d\n}\n" % (pb, queue_delay_us))
    return "\n".join(out)
