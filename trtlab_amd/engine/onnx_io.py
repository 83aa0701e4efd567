"""ONNX model import/export for the trtlab_amd IR.

Import: parses an .onnx file (protobuf wire, onnx_wire.py) into a Graph —
the role the reference delegates to TensorRT's ONNX parser
(examples/ONNX/resnet50/build.py). Supported op set (ResNet-family inference
graphs): Conv, BatchNormalization, Relu, Add, MaxPool, GlobalAveragePool,
Flatten, Gemm, MatMul, Softmax, LayerNormalization, Gelu.

Layout note: ONNX activations are NCHW; the IR is NHWC. Op semantics here
are layout-independent (weights are imported in their canonical ONNX shape
[Cout, Cin, KH, KW] and the engine re-packs), so the importer only rewrites
the input ValueInfo shape NCHW -> NHWC.

Export: serializes an IR graph back to ONNX bytes (round-trip tested;
also a plan-inspection artifact).
"""
from __future__ import annotations

import struct
from typing import Dict, List, Optional

import numpy as np

from trtlab_amd.engine import onnx_wire as w
from trtlab_amd.engine.ir import Graph

# onnx.proto3 field numbers
_MODEL_GRAPH = 7
_GRAPH_NODE, _GRAPH_NAME, _GRAPH_INIT = 1, 2, 5
_GRAPH_INPUT, _GRAPH_OUTPUT = 11, 12
_NODE_INPUT, _NODE_OUTPUT, _NODE_NAME, _NODE_OPTYPE, _NODE_ATTR = 1, 2, 3, 4, 5
_ATTR_NAME, _ATTR_F, _ATTR_I, _ATTR_INTS = 1, 2, 3, 8
_T_DIMS, _T_DTYPE, _T_FLOAT_DATA, _T_INT64_DATA, _T_NAME, _T_RAW = 1, 2, 4, 7, 8, 9
_VI_NAME, _VI_TYPE = 1, 2
_TP_TENSOR = 1
_TT_ELEM, _TT_SHAPE = 1, 2
_TS_DIM = 1
_TD_VALUE = 1

_F32, _F16, _I64, _I32 = 1, 10, 7, 6


# -------------------------------------------------------------------- load
def _parse_tensor(buf: bytes) -> tuple[str, np.ndarray]:
    d = w.fields_dict(buf)
    dims = [w.varint_to_sint64(v) for v in d.get(_T_DIMS, [])]
    dtype = d.get(_T_DTYPE, [_F32])[0]
    name = d.get(_T_NAME, [b""])[0].decode()
    if _T_RAW in d:
        raw = d[_T_RAW][0]
        np_dt = {_F32: np.float32, _F16: np.float16, _I64: np.int64}[dtype]
        arr = np.frombuffer(raw, dtype=np_dt)
    elif _T_FLOAT_DATA in d:
        arr = np.array(
            [struct.unpack("<f", struct.pack("<I", v & 0xFFFFFFFF))[0]
             if isinstance(v, int) else v
             for v in d[_T_FLOAT_DATA]], dtype=np.float32)
    elif _T_INT64_DATA in d:
        arr = np.array([w.varint_to_sint64(v) for v in d[_T_INT64_DATA]],
                       dtype=np.int64)
    else:
        arr = np.zeros(0, np.float32)
    return name, arr.reshape(dims) if dims else arr


def _parse_attrs(bufs: List[bytes]) -> Dict[str, object]:
    out: Dict[str, object] = {}
    for b in bufs:
        d = w.fields_dict(b)
        name = d[_ATTR_NAME][0].decode()
        if _ATTR_INTS in d:
            vals = []
            for v in d[_ATTR_INTS]:
                if isinstance(v, bytes):  # packed
                    vals.extend(w.decode_packed_varints(v))
                else:
                    vals.append(w.varint_to_sint64(v))
            out[name] = vals
        elif _ATTR_I in d:
            out[name] = w.varint_to_sint64(d[_ATTR_I][0])
        elif _ATTR_F in d:
            out[name] = struct.unpack("<f", struct.pack("<i", d[_ATTR_F][0]))[0]
    return out


def _parse_vi_shape(buf: bytes) -> tuple[str, List[int], int]:
    d = w.fields_dict(buf)
    name = d[_VI_NAME][0].decode()
    dims: List[int] = []
    elem = _F32
    if _VI_TYPE in d:
        tp = w.fields_dict(d[_VI_TYPE][0])
        if _TP_TENSOR in tp:
            tt = w.fields_dict(tp[_TP_TENSOR][0])
            elem = tt.get(_TT_ELEM, [_F32])[0]
            if _TT_SHAPE in tt:
                for dim_buf in w.fields_dict(tt[_TT_SHAPE][0]).get(_TS_DIM, []):
                    dd = w.fields_dict(dim_buf)
                    dims.append(w.varint_to_sint64(dd.get(_TD_VALUE, [0])[0]))
    return name, dims, elem


def import_onnx(data: bytes, batch: Optional[int] = None,
                name: str = "onnx_model") -> Graph:
    model = w.fields_dict(data)
    graph_buf = model[_MODEL_GRAPH][0]
    gd = w.fields_dict(graph_buf)

    inits: Dict[str, np.ndarray] = {}
    for t in gd.get(_GRAPH_INIT, []):
        nm, arr = _parse_tensor(t)
        inits[nm] = arr

    g = Graph(name)
    # graph input (the one without an initializer)
    input_name = None
    for vi in gd.get(_GRAPH_INPUT, []):
        nm, dims, elem = _parse_vi_shape(vi)
        if nm in inits:
            continue
        dt = {_F32: "f32", _F16: "f16", _I32: "i32", _I64: "i32"}.get(
            elem, "f16")
        if dt == "f32":
            dt = "f16"  # fp32 graph inputs feed the fp16 compute path
        if len(dims) == 4:
            n, c, h, ww = dims  # NCHW -> IR NHWC
            if batch:
                n = batch
            input_name = g.input((n, h, ww, c), name=nm)
        else:  # 2-D (features) / 1-D (token ids): layout-free
            if batch and dims:
                dims = [batch] + list(dims[1:])
            input_name = g.input(tuple(dims), name=nm, dtype=dt)
    assert input_name is not None, "no graph input found"

    # name remapping: ONNX tensor name -> IR tensor name
    remap: Dict[str, str] = {input_name: input_name}

    for nbuf in gd.get(_GRAPH_NODE, []):
        nd = w.fields_dict(nbuf)
        op = nd[_NODE_OPTYPE][0].decode()
        ins = [b.decode() for b in nd.get(_NODE_INPUT, [])]
        outs = [b.decode() for b in nd.get(_NODE_OUTPUT, [])]
        attrs = _parse_attrs(nd.get(_NODE_ATTR, []))
        x = remap.get(ins[0], ins[0]) if ins else None

        if op == "Conv":
            wt = inits[ins[1]].astype(np.float32)
            strides = attrs.get("strides", [1, 1])
            pads = attrs.get("pads", [0, 0, 0, 0])
            out = g.conv2d(x, wt, stride=int(strides[0]),
                           padding=int(pads[0]))
            if len(ins) > 2:  # conv bias -> fold into a batchnorm-less bias
                b = inits[ins[2]].astype(np.float32)
                cout = wt.shape[0]
                out = g.batchnorm(out, gamma=np.ones(cout, np.float32),
                                  beta=b, mean=np.zeros(cout, np.float32),
                                  var=np.ones(cout, np.float32) - 1e-5)
        elif op == "BatchNormalization":
            gamma, beta, mean, var = (inits[ins[k]].astype(np.float32)
                                      for k in (1, 2, 3, 4))
            out = g.batchnorm(x, gamma=gamma, beta=beta, mean=mean, var=var,
                              eps=float(attrs.get("epsilon", 1e-5)))
        elif op == "Relu":
            out = g.relu(x)
        elif op == "Gelu":
            out = g.gelu(x)
        elif op == "Add":
            other = remap.get(ins[1], ins[1])
            if other in inits and other not in g.tensors:
                # constant operand (e.g. ViT position embeddings):
                # materialize it as a device-resident constant tensor
                other = g.constant(inits[other].astype(np.float32),
                                   name=other + "_const")
            out = g.add(x, other)
        elif op == "AveragePool":
            ks = attrs.get("kernel_shape", [2, 2])
            strides = attrs.get("strides", [1, 1])
            pads = attrs.get("pads", [0, 0, 0, 0])
            out = g.avgpool(x, kernel=int(ks[0]), stride=int(strides[0]),
                            padding=int(pads[0]))
        elif op == "MaxPool":
            ks = attrs.get("kernel_shape", [2, 2])
            strides = attrs.get("strides", [1, 1])
            pads = attrs.get("pads", [0, 0, 0, 0])
            out = g.maxpool(x, kernel=int(ks[0]), stride=int(strides[0]),
                            padding=int(pads[0]))
        elif op == "GlobalAveragePool":
            out = g.global_avgpool(x)
        elif op in ("Flatten", "Reshape", "Squeeze", "Unsqueeze"):
            # shape plumbing. When the target shape is a static
            # initializer and actually differs, emit a zero-copy view
            # (ViT token flatten / pool reshape); otherwise pass through
            # (torchvision GAP->Gemm heads where the IR is already 2-D).
            tgt = None
            if op == "Reshape" and len(ins) > 1 and ins[1] in inits:
                tgt = [int(v) for v in
                       np.asarray(inits[ins[1]]).reshape(-1)]
            cur = list(g.tensors[x].shape)
            if tgt and tgt != cur and all(d > 0 for d in tgt):
                out = g.view(x, tuple(tgt))
            else:
                out = x
        elif op in ("Identity", "Dropout"):
            out = x  # inference mode: both are pass-through
        elif op == "Clip":
            # opset<11: attrs; opset>=11: optional min/max initializer inputs
            mn = attrs.get("min")
            mx = attrs.get("max")
            if mn is None and len(ins) > 1 and ins[1] and ins[1] in inits:
                mn = float(np.asarray(inits[ins[1]]).reshape(-1)[0])
            if mx is None and len(ins) > 2 and ins[2] and ins[2] in inits:
                mx = float(np.asarray(inits[ins[2]]).reshape(-1)[0])
            if (mn is None or mn == 0.0) and (mx is None or mx >= 3e38):
                out = g.relu(x)  # ReLU-equivalent: fusable epilogue
            else:  # general bounds (ReLU6 etc.) -> dedicated clip kernel
                out = g.clip(x, -3e38 if mn is None else mn,
                             3e38 if mx is None else mx)
        elif op == "Sum":
            # N-ary Sum -> chain of adds (2 inputs = plain Add)
            out = remap.get(ins[0], ins[0])
            for extra in ins[1:]:
                out = g.add(out, remap.get(extra, extra))
        elif op == "Concat":
            a = attrs.get("axis", 1)
            axis = int(a[0] if isinstance(a, (list, tuple)) else a)
            srcs = [remap.get(i, i) for i in ins]
            rank = len(g.tensors[srcs[0]].shape)
            # ONNX graphs are NCHW; the IR is NHWC — channel concat
            # (axis=1, rank 4) is last-axis concat here; last-axis concat
            # (features, rank 2) maps directly
            if not (axis in (1, -1, rank - 1) or
                    (rank == 4 and axis == 1)):
                raise ValueError(f"ONNX Concat axis {axis} not supported")
            out = g.concat(srcs)
        elif op == "Transpose":
            perm = [int(v) for v in attrs.get("perm", [])]
            rank = len(g.tensors[x].shape)
            if rank == 2 and (perm == [1, 0] or not perm):
                out = g.transpose2d(x)
            elif perm == list(range(rank)):
                out = x  # identity permutation
            elif rank == 4 and perm in ([0, 2, 3, 1], [0, 3, 1, 2]):
                # NCHW<->NHWC annotations: our activations are ALREADY
                # NHWC, so a single layout flip is a no-op view; chains
                # (flip + flip back) cancel
                out = x
            else:
                raise ValueError(f"ONNX Transpose perm {perm} not supported")
        elif op in ("Gemm", "MatMul"):
            wt = inits[ins[1]].astype(np.float32)
            if op == "MatMul" or not attrs.get("transB", 0):
                wt = wt.T  # IR gemm takes [out, in]
            bias = inits[ins[2]].astype(np.float32) if len(ins) > 2 else None
            out = g.gemm(x, np.ascontiguousarray(wt), bias)
        elif op == "Softmax":
            out = g.softmax(x)
        elif op == "LayerNormalization":
            out = g.layernorm(x, inits[ins[1]].astype(np.float32),
                              inits[ins[2]].astype(np.float32),
                              eps=float(attrs.get("epsilon", 1e-5)))
        elif op == "TrtlabEmbedding":
            # custom-domain round-trip op: [ids, tok, pos, (seg), (segids)]
            tables = [i for i in ins[1:] if i in inits]
            tensors = [i for i in ins[1:] if i not in inits]
            tok = inits[tables[0]].astype(np.float32)
            pos = inits[tables[1]].astype(np.float32)
            seg = (inits[tables[2]].astype(np.float32)
                   if len(tables) > 2 else None)
            segids = remap.get(tensors[0], tensors[0]) if tensors else None
            out = g.embedding(x, tok, pos, seg_table=seg, segids=segids)
        elif op == "TrtlabAttention":
            out = g.attention(x, heads=int(attrs["heads"]),
                              seq=int(attrs["seq"]),
                              causal=bool(attrs.get("causal", 0)),
                              varlen=bool(attrs.get("varlen", 0)),
                              pad_id=int(attrs.get("pad_id", 0)))
        elif op == "TrtlabRMSNorm":
            out = g.rmsnorm(x, inits[ins[1]].astype(np.float32),
                            eps=float(attrs.get("epsilon", 1e-5)))
        elif op == "TrtlabSiluMul":
            out = g.silu_mul(x, remap.get(ins[1], ins[1]))
        elif op == "TrtlabRope":
            out = g.rope(x, heads=int(attrs["heads"]),
                         seq=int(attrs["seq"]),
                         theta=float(attrs.get("theta", 10000.0)))
        else:
            raise ValueError(f"ONNX op {op} not supported by the importer")
        remap[outs[0]] = out
    return g


def load_onnx(path: str, batch: Optional[int] = None) -> Graph:
    with open(path, "rb") as f:
        return import_onnx(f.read(), batch=batch)


# ------------------------------------------------------------------ export
def _tensor_bytes(name: str, arr: np.ndarray) -> bytes:
    out = b"".join(w.f_varint(_T_DIMS, d) for d in arr.shape)
    out += w.f_varint(_T_DTYPE, _F32)
    out += w.f_string(_T_NAME, name)
    out += w.f_bytes(_T_RAW, np.ascontiguousarray(arr, np.float32).tobytes())
    return out


def _attr_ints(name: str, vals: List[int]) -> bytes:
    body = w.f_string(_ATTR_NAME, name)
    for v in vals:
        body += w.f_varint(_ATTR_INTS, v)
    body += w.f_varint(20, 7)  # AttributeProto.type = INTS
    return w.f_bytes(_NODE_ATTR, body)


def _attr_f(name: str, v: float) -> bytes:
    body = w.f_string(_ATTR_NAME, name) + w.f_float(_ATTR_F, v)
    body += w.f_varint(20, 1)  # FLOAT
    return w.f_bytes(_NODE_ATTR, body)


def _attr_i(name: str, v: int) -> bytes:
    body = w.f_string(_ATTR_NAME, name) + w.f_varint(_ATTR_I, v)
    body += w.f_varint(20, 2)  # INT
    return w.f_bytes(_NODE_ATTR, body)


def _node(op: str, ins: List[str], outs: List[str], *attrs: bytes) -> bytes:
    body = b"".join(w.f_string(_NODE_INPUT, i) for i in ins)
    body += b"".join(w.f_string(_NODE_OUTPUT, o) for o in outs)
    body += w.f_string(_NODE_OPTYPE, op)
    body += b"".join(attrs)
    return w.f_bytes(_GRAPH_NODE, body)


def _value_info(name: str, dims: List[int], elem: int = _F32) -> bytes:
    dim_bufs = b"".join(
        w.f_bytes(_TS_DIM, w.f_varint(_TD_VALUE, d)) for d in dims)
    shape = w.f_bytes(_TT_SHAPE, dim_bufs)
    tt = w.f_varint(_TT_ELEM, elem) + shape
    tp = w.f_bytes(_TP_TENSOR, tt)
    return w.f_string(_VI_NAME, name) + w.f_bytes(_VI_TYPE, tp)


def export_onnx(g: Graph) -> bytes:
    """Serialize the (pre-fusion) IR graph to ONNX bytes."""
    nodes = b""
    inits = b""
    init_ct = 0

    def add_init(arr: np.ndarray) -> str:
        nonlocal inits, init_ct
        nm = f"w{init_ct}"
        init_ct += 1
        inits += w.f_bytes(_GRAPH_INIT, _tensor_bytes(nm, arr))
        return nm

    for n in g.nodes:
        if n.kind == "input":
            continue
        if n.kind == "conv2d":
            wn = add_init(n.attrs["weight"])
            p = n.attrs["padding"]
            nodes += _node("Conv", [n.inputs[0], wn], [n.output],
                           _attr_ints("strides", [n.attrs["stride"]] * 2),
                           _attr_ints("pads", [p] * 4),
                           _attr_ints("kernel_shape",
                                      list(n.attrs["weight"].shape[2:])))
        elif n.kind == "batchnorm":
            a = n.attrs
            names = [add_init(a[k]) for k in ("gamma", "beta", "mean", "var")]
            nodes += _node("BatchNormalization", [n.inputs[0]] + names,
                           [n.output], _attr_f("epsilon", a["eps"]))
        elif n.kind == "relu":
            nodes += _node("Relu", [n.inputs[0]], [n.output])
        elif n.kind == "gelu":
            nodes += _node("Gelu", [n.inputs[0]], [n.output])
        elif n.kind == "add":
            nodes += _node("Add", list(n.inputs), [n.output])
        elif n.kind == "maxpool":
            a = n.attrs
            nodes += _node("MaxPool", [n.inputs[0]], [n.output],
                           _attr_ints("kernel_shape", [a["kernel"]] * 2),
                           _attr_ints("strides", [a["stride"]] * 2),
                           _attr_ints("pads", [a["padding"]] * 4))
        elif n.kind == "gavgpool":
            nodes += _node("GlobalAveragePool", [n.inputs[0]], [n.output])
        elif n.kind == "gemm":
            wn = add_init(n.attrs["weight"])
            ins = [n.inputs[0], wn]
            if n.attrs.get("bias") is not None:
                ins.append(add_init(n.attrs["bias"]))
            nodes += _node("Gemm", ins, [n.output], _attr_i("transB", 1))
        elif n.kind == "softmax":
            nodes += _node("Softmax", [n.inputs[0]], [n.output])
        elif n.kind == "layernorm":
            a = n.attrs
            nodes += _node("LayerNormalization",
                           [n.inputs[0], add_init(a["gamma"]),
                            add_init(a["beta"])], [n.output],
                           _attr_f("epsilon", a["eps"]))
        elif n.kind == "avgpool":
            a = n.attrs
            nodes += _node("AveragePool", [n.inputs[0]], [n.output],
                           _attr_ints("kernel_shape", [a["kernel"]] * 2),
                           _attr_ints("strides", [a["stride"]] * 2),
                           _attr_ints("pads", [a["padding"]] * 4))
        elif n.kind == "clip":
            nodes += _node("Clip", [n.inputs[0]], [n.output],
                           _attr_f("min", n.attrs["mn"]),
                           _attr_f("max", n.attrs["mx"]))
        elif n.kind == "transpose2d":
            nodes += _node("Transpose", [n.inputs[0]], [n.output],
                           _attr_ints("perm", [1, 0]))
        elif n.kind == "concat":
            rank = len(g.tensors[n.inputs[0]].shape)
            nodes += _node("Concat", list(n.inputs), [n.output],
                           _attr_i("axis", rank - 1))
        elif n.kind == "add_layernorm":
            # decompose to standard ops: Add + LayerNormalization
            a = n.attrs
            mid = n.output + "_sum"
            nodes += _node("Add", list(n.inputs), [mid])
            nodes += _node("LayerNormalization",
                           [mid, add_init(a["gamma"]), add_init(a["beta"])],
                           [n.output], _attr_f("epsilon", a["eps"]))
        elif n.kind == "embedding":
            # custom-domain op with the tables as initializers (transformer
            # round-trip; standard ONNX would need Gather+Add chains the IR
            # has no generic ops for)
            a = n.attrs
            ins = [n.inputs[0], add_init(a["tok"]), add_init(a["pos"])]
            if a.get("seg") is not None:
                ins.append(add_init(a["seg"]))
            nodes += _node("TrtlabEmbedding", ins + list(n.inputs[1:]),
                           [n.output], _attr_i("seq", a["seq"]))
        elif n.kind == "attention":
            a = n.attrs
            nodes += _node("TrtlabAttention", [n.inputs[0]], [n.output],
                           _attr_i("heads", a["heads"]),
                           _attr_i("seq", a["seq"]),
                           _attr_i("causal", 1 if a.get("causal") else 0),
                           _attr_i("varlen", 1 if a.get("varlen") else 0),
                           _attr_i("pad_id", a.get("pad_id", 0)))
        elif n.kind == "view":
            # Reshape with the target shape as an int64 initializer
            shp = np.asarray(g.tensors[n.output].shape, np.int64)
            nodes += _node("Reshape", [n.inputs[0], add_init(shp)],
                           [n.output])
        elif n.kind == "constant":
            # the constant tensor becomes a graph initializer named like
            # the node's output (standard ONNX constant-folding shape)
            inits += w.f_bytes(_GRAPH_INIT,
                               _tensor_bytes(n.output, n.attrs["value"]))
        elif n.kind == "rmsnorm":
            nodes += _node("TrtlabRMSNorm",
                           [n.inputs[0], add_init(n.attrs["gamma"])],
                           [n.output], _attr_f("epsilon", n.attrs["eps"]))
        elif n.kind == "silu_mul":
            nodes += _node("TrtlabSiluMul", list(n.inputs), [n.output])
        elif n.kind == "rope":
            a = n.attrs
            nodes += _node("TrtlabRope", [n.inputs[0]], [n.output],
                           _attr_i("heads", a["heads"]),
                           _attr_i("seq", a["seq"]),
                           _attr_f("theta", a["theta"]))
        else:
            raise ValueError(f"export: unsupported node kind {n.kind}")

    shape = list(g.tensors[g.input_name].shape)
    if len(shape) == 4:
        nb, h, ww, c = shape
        shape = [nb, c, h, ww]  # IR NHWC -> ONNX NCHW
    in_vis = b""
    for nm in (g.input_names or [g.input_name]):
        s_ = list(g.tensors[nm].shape)
        if len(s_) == 4:
            s_ = [s_[0], s_[3], s_[1], s_[2]]
        elem = _I32 if g.tensors[nm].dtype == "i32" else _F32
        in_vis += w.f_bytes(_GRAPH_INPUT, _value_info(nm, s_, elem))
    out_vi = w.f_bytes(_GRAPH_OUTPUT, _value_info(g.output_name, []))
    graph = nodes + w.f_string(_GRAPH_NAME, g.name) + inits + in_vis + out_vi
    model = w.f_varint(1, 8)  # ir_version
    model += w.f_bytes(_MODEL_GRAPH, graph)
    return model
