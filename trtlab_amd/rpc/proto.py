"""Protobuf message definitions built at import time via dynamic descriptors
(no protoc in the offline image). Wire-compatible with a .proto of the same
shape — mirrors the reference's testing.proto / nvidia_inference.proto
surfaces.

    package trtlab;
    message EchoRequest   { string message = 1; int64 tag = 2; }
    message EchoResponse  { string message = 1; int64 tag = 2; }
    message HealthRequest { }
    message HealthResponse{ bool ready = 1; string status = 2; }
    message InferRequest  { string model = 1; bytes input = 2;
                            repeated int64 shape = 3; string dtype = 4;
                            int64 batch_id = 5;
                            string shm_name = 6; int64 shm_size = 7; }
    message InferResponse { bytes output = 1; repeated int64 shape = 2;
                            string dtype = 3; int64 batch_id = 4;
                            float compute_ms = 5; float request_ms = 6; }
"""
from __future__ import annotations

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

_pool = descriptor_pool.Default()

_f = descriptor_pb2.FileDescriptorProto()
_f.name = "trtlab_amd/rpc/trtlab.proto"
_f.package = "trtlab"
_f.syntax = "proto3"

_T = descriptor_pb2.FieldDescriptorProto


def _msg(name, fields):
    m = _f.message_type.add()
    m.name = name
    for num, (fname, ftype, repeated) in enumerate(fields, start=1):
        fd = m.field.add()
        fd.name = fname
        fd.number = num
        fd.type = ftype
        fd.label = (_T.LABEL_REPEATED if repeated else _T.LABEL_OPTIONAL)


_msg("EchoRequest", [("message", _T.TYPE_STRING, False),
                     ("tag", _T.TYPE_INT64, False)])
_msg("EchoResponse", [("message", _T.TYPE_STRING, False),
                      ("tag", _T.TYPE_INT64, False)])
_msg("HealthRequest", [])
_msg("HealthResponse", [("ready", _T.TYPE_BOOL, False),
                        ("status", _T.TYPE_STRING, False)])
# Named tensor for multi-binding models (reference Bindings carve N
# addresses per model, bindings.h:60-120; TRTIS InferRequestHeader names
# its inputs the same way).
_msg("NamedTensor", [("name", _T.TYPE_STRING, False),
                     ("data", _T.TYPE_BYTES, False),
                     ("shape", _T.TYPE_INT64, True),
                     ("dtype", _T.TYPE_STRING, False)])


def _msg_field(m_name, fname, num, type_name):
    m = next(m for m in _f.message_type if m.name == m_name)
    fd = m.field.add()
    fd.name = fname
    fd.number = num
    fd.type = _T.TYPE_MESSAGE
    fd.type_name = f".trtlab.{type_name}"
    fd.label = _T.LABEL_REPEATED


_msg("InferRequest", [("model", _T.TYPE_STRING, False),
                      ("input", _T.TYPE_BYTES, False),
                      ("shape", _T.TYPE_INT64, True),
                      ("dtype", _T.TYPE_STRING, False),
                      ("batch_id", _T.TYPE_INT64, False),
                      # zero-copy local input: POSIX shared memory name
                      # (reference SysV SharedMemoryService,
                      # testing.proto:37-48)
                      ("shm_name", _T.TYPE_STRING, False),
                      ("shm_size", _T.TYPE_INT64, False)])
_msg("InferResponse", [("output", _T.TYPE_BYTES, False),
                       ("shape", _T.TYPE_INT64, True),
                       ("dtype", _T.TYPE_STRING, False),
                       ("batch_id", _T.TYPE_INT64, False),
                       ("compute_ms", _T.TYPE_FLOAT, False),
                       ("request_ms", _T.TYPE_FLOAT, False)])
# field 8: repeated NamedTensor inputs; field 7: repeated NamedTensor outputs
_msg_field("InferRequest", "inputs", 8, "NamedTensor")
_msg_field("InferResponse", "outputs", 7, "NamedTensor")

_file_desc = _pool.Add(_f)


def _cls(name):
    return message_factory.GetMessageClass(_pool.FindMessageTypeByName(f"trtlab.{name}"))


EchoRequest = _cls("EchoRequest")
NamedTensor = _cls("NamedTensor")
EchoResponse = _cls("EchoResponse")
HealthRequest = _cls("HealthRequest")
HealthResponse = _cls("HealthResponse")
InferRequest = _cls("InferRequest")
InferResponse = _cls("InferResponse")
