"""EnginePlan serialization: the compiled-plan cache (reference analogue:
serialized TensorRT .engine files, runtime.h:63 read_engine_file — here the
plan is explicit data: op dicts + weight blob + layout, stored as one .npz).
"""
from __future__ import annotations

import json
from typing import Union

import numpy as np

from trtlab_amd.engine.planner import EnginePlan, ExecOp

_FORMAT_VERSION = 1


def save_plan(plan: EnginePlan, path: str) -> None:
    meta = dict(
        version=_FORMAT_VERSION,
        name=plan.name,
        ops=plan.ops,
        arena_bytes=plan.arena_bytes,
        offsets=plan.offsets,
        input_name=plan.input_name,
        input_off=plan.input_off,
        input_bytes=plan.input_bytes,
        input_shape=list(plan.input_shape),
        output_name=plan.output_name,
        output_off=plan.output_off,
        output_bytes=plan.output_bytes,
        output_shape=list(plan.output_shape),
        dtype=plan.dtype,
        shapes={k: list(v) for k, v in plan.shapes.items()},
        input_dtype=plan.input_dtype,
        inputs=[{**b, "shape": list(b["shape"])} for b in plan.inputs],
        outputs=[{**b, "shape": list(b["shape"])} for b in plan.outputs],
        exec_meta=[dict(kind=o.kind, name=o.name, inputs=o.inputs,
                        output=o.output,
                        params={k: v for k, v in o.params.items()
                                if not isinstance(v, np.ndarray)})
                   for o in plan.exec_ops],
    )
    arrays = {"weights": plan.weights}
    for i, o in enumerate(plan.exec_ops):
        if o.w is not None:
            arrays[f"w{i}"] = o.w
        if o.scale is not None:
            arrays[f"s{i}"] = o.scale
        if o.bias is not None:
            arrays[f"b{i}"] = o.bias
        for k, v in o.params.items():  # ndarray params (e.g. MX wt scales)
            if isinstance(v, np.ndarray):
                arrays[f"p{i}_{k}"] = v
    # write through a file object so the exact path is honored
    # (np.savez appends .npz to bare string paths)
    with open(path, "wb") as f:
        np.savez_compressed(f, meta=np.frombuffer(
            json.dumps(meta).encode(), dtype=np.uint8), **arrays)


def load_plan(path: str) -> EnginePlan:
    z = np.load(path, allow_pickle=False)
    meta = json.loads(bytes(z["meta"]).decode())
    if meta["version"] != _FORMAT_VERSION:
        raise ValueError(f"plan format version {meta['version']} unsupported")
    exec_ops = []
    for i, em in enumerate(meta["exec_meta"]):
        op = ExecOp(em["kind"], em["name"], list(em["inputs"]), em["output"],
                    dict(em["params"]))
        if f"w{i}" in z:
            op.w = z[f"w{i}"]
        if f"s{i}" in z:
            op.scale = z[f"s{i}"]
        if f"b{i}" in z:
            op.bias = z[f"b{i}"]
        for key in z.files:
            if key.startswith(f"p{i}_"):
                op.params[key[len(f"p{i}_"):]] = z[key]
        exec_ops.append(op)
    return EnginePlan(
        name=meta["name"],
        ops=meta["ops"],
        exec_ops=exec_ops,
        weights=z["weights"],
        arena_bytes=meta["arena_bytes"],
        offsets=meta["offsets"],
        input_name=meta["input_name"],
        input_off=meta["input_off"],
        input_bytes=meta["input_bytes"],
        input_shape=tuple(meta["input_shape"]),
        output_name=meta["output_name"],
        output_off=meta["output_off"],
        output_bytes=meta["output_bytes"],
        output_shape=tuple(meta["output_shape"]),
        dtype=meta["dtype"],
        shapes={k: tuple(v) for k, v in meta["shapes"].items()},
        input_dtype=meta.get("input_dtype", "f16"),
        inputs=[{**b, "shape": tuple(b["shape"])}
                for b in meta.get("inputs", [])],
        outputs=[{**b, "shape": tuple(b["shape"])}
                 for b in meta.get("outputs", [])],
    )
