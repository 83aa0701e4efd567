"""Model IR: a small typed graph of inference ops, NHWC layout, fp16 compute.

The IR is built by model builders (trtlab_amd.models) or the ONNX importer,
then lowered by the Planner (fusion + memory planning) into an executable
plan. Replaces the reference's reliance on nvinfer1::ICudaEngine
(trtlab/tensorrt/model.h:17) with an explicit, inspectable graph.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple

import numpy as np


@dataclass
class TensorSpec:
    name: str
    shape: Tuple[int, ...]  # NHWC for 4-D activations, [M, K] for 2-D
    dtype: str = "f16"

    @property
    def numel(self) -> int:
        n = 1
        for s in self.shape:
            n *= int(s)
        return n

    @property
    def nbytes(self) -> int:
        itemsize = {"f16": 2, "bf16": 2, "f32": 4, "i32": 4, "i8": 1}[self.dtype]
        return self.numel * itemsize


@dataclass
class Node:
    kind: str                      # conv2d | batchnorm | relu | gelu | add |
    #                                maxpool | gavgpool | flatten | gemm |
    #                                softmax | layernorm | add_layernorm |
    #                                attention | input
    name: str
    inputs: List[str]
    output: str
    attrs: Dict[str, Any] = field(default_factory=dict)


class Graph:
    """Topologically-ordered op list with named tensors."""

    def __init__(self, name: str = "model"):
        self.name = name
        self.nodes: List[Node] = []
        self.tensors: Dict[str, TensorSpec] = {}
        # N-binding I/O (reference bindings.h:60-120 carves N host+device
        # addresses per model): input_names in declaration order;
        # output_names pinned via mark_output() (default: the last op's
        # output). input_name/output_name remain the primary bindings.
        self.input_names: List[str] = []
        self.output_names: List[str] = []
        self.input_name: Optional[str] = None
        self.output_name: Optional[str] = None
        self._ctr = 0

    # ---------------------------------------------------------- builders
    def _fresh(self, prefix: str) -> str:
        self._ctr += 1
        return f"{prefix}_{self._ctr}"

    def add_tensor(self, name: str, shape: Tuple[int, ...], dtype: str = "f16") -> str:
        self.tensors[name] = TensorSpec(name, tuple(int(s) for s in shape), dtype)
        return name

    def input(self, shape: Tuple[int, ...], name: str = "input",
              dtype: str = "f16") -> str:
        self.add_tensor(name, shape, dtype)
        self.nodes.append(Node("input", name, [], name))
        if self.input_name is None:
            self.input_name = name  # primary binding = first declared
        self.input_names.append(name)
        return name

    def mark_output(self, name: str) -> str:
        """Pin a tensor as an engine output binding (in addition to the
        default last-op output). Order of mark_output calls = binding
        order; the primary output stays output_name unless marked first."""
        if name not in self.tensors:
            raise KeyError(f"mark_output: unknown tensor {name}")
        if name not in self.output_names:
            self.output_names.append(name)
        return name

    def _emit(self, kind: str, inputs: List[str], out_shape: Tuple[int, ...],
              attrs: Dict[str, Any], name: Optional[str] = None) -> str:
        name = name or self._fresh(kind)
        out = self.add_tensor(name, out_shape)
        self.nodes.append(Node(kind, name, list(inputs), out, attrs))
        self.output_name = out
        return out

    def conv2d(self, x: str, weight: np.ndarray, stride: int = 1,
               padding: int = 0, name: Optional[str] = None) -> str:
        n, h, w, c = self.tensors[x].shape
        cout, cin, kh, kw = weight.shape
        assert cin == c, f"conv2d: Cin {cin} != input C {c}"
        oh = (h + 2 * padding - kh) // stride + 1
        ow = (w + 2 * padding - kw) // stride + 1
        return self._emit("conv2d", [x], (n, oh, ow, cout),
                          dict(weight=np.asarray(weight, np.float32),
                               stride=stride, padding=padding), name)

    def batchnorm(self, x: str, gamma, beta, mean, var, eps: float = 1e-5,
                  name: Optional[str] = None) -> str:
        return self._emit("batchnorm", [x], self.tensors[x].shape,
                          dict(gamma=np.asarray(gamma, np.float32),
                               beta=np.asarray(beta, np.float32),
                               mean=np.asarray(mean, np.float32),
                               var=np.asarray(var, np.float32), eps=eps), name)

    def relu(self, x: str, name: Optional[str] = None) -> str:
        return self._emit("relu", [x], self.tensors[x].shape, {}, name)

    def gelu(self, x: str, name: Optional[str] = None) -> str:
        return self._emit("gelu", [x], self.tensors[x].shape, {}, name)

    def add(self, a: str, b: str, name: Optional[str] = None) -> str:
        return self._emit("add", [a, b], self.tensors[a].shape, {}, name)

    def maxpool(self, x: str, kernel: int, stride: int, padding: int = 0,
                name: Optional[str] = None) -> str:
        n, h, w, c = self.tensors[x].shape
        oh = (h + 2 * padding - kernel) // stride + 1
        ow = (w + 2 * padding - kernel) // stride + 1
        return self._emit("maxpool", [x], (n, oh, ow, c),
                          dict(kernel=kernel, stride=stride, padding=padding),
                          name)

    def avgpool(self, x: str, kernel: int, stride: int, padding: int = 0,
                name: Optional[str] = None) -> str:
        n, h, w, c = self.tensors[x].shape
        oh = (h + 2 * padding - kernel) // stride + 1
        ow = (w + 2 * padding - kernel) // stride + 1
        return self._emit("avgpool", [x], (n, oh, ow, c),
                          dict(kernel=kernel, stride=stride, padding=padding),
                          name)

    def global_avgpool(self, x: str, name: Optional[str] = None) -> str:
        n, h, w, c = self.tensors[x].shape
        return self._emit("gavgpool", [x], (n, c), dict(hw=h * w), name)

    def gemm(self, x: str, weight: np.ndarray, bias: Optional[np.ndarray] = None,
             name: Optional[str] = None) -> str:
        m, k = self.tensors[x].shape
        nout, kin = weight.shape
        assert kin == k, f"gemm: K {kin} != input {k}"
        return self._emit("gemm", [x], (m, nout),
                          dict(weight=np.asarray(weight, np.float32),
                               bias=None if bias is None else np.asarray(bias, np.float32)),
                          name)

    def softmax(self, x: str, name: Optional[str] = None) -> str:
        return self._emit("softmax", [x], self.tensors[x].shape, {}, name)

    def layernorm(self, x: str, gamma, beta, eps: float = 1e-5,
                  name: Optional[str] = None) -> str:
        return self._emit("layernorm", [x], self.tensors[x].shape,
                          dict(gamma=np.asarray(gamma, np.float32),
                               beta=np.asarray(beta, np.float32), eps=eps),
                          name)

    def add_layernorm(self, x: str, res: str, gamma, beta, eps: float = 1e-5,
                      name: Optional[str] = None) -> str:
        return self._emit("add_layernorm", [x, res], self.tensors[x].shape,
                          dict(gamma=np.asarray(gamma, np.float32),
                               beta=np.asarray(beta, np.float32), eps=eps),
                          name)

    def embedding(self, ids: str, tok_table: np.ndarray,
                  pos_table: np.ndarray, seg_table=None,
                  segids: Optional[str] = None,
                  name: Optional[str] = None) -> str:
        """out[m] = tok[ids[m]] + pos[m % S] (+ seg[segids[m]]). segids:
        optional i32 input-tensor name (BERT token_type_ids binding)."""
        m = self.tensors[ids].shape[0]
        v, h = tok_table.shape
        s_, h2 = pos_table.shape
        assert h == h2
        ins = [ids] if segids is None else [ids, segids]
        return self._emit("embedding", ins, (m, h),
                          dict(tok=np.asarray(tok_table, np.float32),
                               pos=np.asarray(pos_table, np.float32),
                               seg=None if seg_table is None
                               else np.asarray(seg_table, np.float32),
                               seq=s_), name)

    def rmsnorm(self, x: str, gamma, eps: float = 1e-5,
                name: Optional[str] = None) -> str:
        """LLaMA norm: x / rms(x) * gamma (no mean subtraction/beta)."""
        return self._emit("rmsnorm", [x], self.tensors[x].shape,
                          dict(gamma=np.asarray(gamma, np.float32),
                               eps=eps), name)

    def silu_mul(self, a: str, b: str, name: Optional[str] = None) -> str:
        """SwiGLU gate: silu(a) * b (LLaMA FFN)."""
        assert self.tensors[a].shape == self.tensors[b].shape
        return self._emit("silu_mul", [a, b], self.tensors[a].shape, {},
                          name)

    def rope(self, qkv: str, heads: int, seq: int, theta: float = 10000.0,
             name: Optional[str] = None) -> str:
        """Rotary position embedding applied to the q/k blocks of fused
        qkv rows (in place — the planner aliases output and input in the
        arena). LLaMA-style half-split rotation, angle base `theta`."""
        return self._emit("rope", [qkv], self.tensors[qkv].shape,
                          dict(heads=heads, seq=seq, theta=float(theta)),
                          name)

    def view(self, x: str, shape: Tuple[int, ...],
             name: Optional[str] = None) -> str:
        """Zero-copy reshape: same row-major bytes under a new shape
        (e.g. NHWC patch grid [B, g, g, H] -> token rows [B*g*g, H]).
        Lowered to an arena alias — no kernel, no copy."""
        src = self.tensors[x]
        n = 1
        for d in shape:
            n *= int(d)
        assert n == src.numel, f"view: numel {n} != {src.numel}"
        return self._emit("view", [x], tuple(int(d) for d in shape), {},
                          name)

    def constant(self, value: np.ndarray,
                 name: Optional[str] = None) -> str:
        """Device-resident constant tensor (packed into the weight blob,
        copied into its arena slot inside the captured graph) — e.g. ViT
        position embeddings added to the patch tokens."""
        value = np.asarray(value, np.float32)
        return self._emit("constant", [], value.shape, dict(value=value),
                          name)

    def clip(self, x: str, mn: float, mx: float,
             name: Optional[str] = None) -> str:
        """out = min(max(x, mn), mx) (ONNX Clip with arbitrary bounds)."""
        return self._emit("clip", [x], self.tensors[x].shape,
                          dict(mn=float(mn), mx=float(mx)), name)

    def transpose2d(self, x: str, name: Optional[str] = None) -> str:
        """out[N][M] = x[M][N]^T (2-D only; the general layout kernel)."""
        m, n = self.tensors[x].shape
        return self._emit("transpose2d", [x], (n, m), {}, name)

    def concat(self, xs: List[str], name: Optional[str] = None) -> str:
        """Concatenate along the LAST axis (channels in NHWC / features in
        2-D) — the common inception/densenet pattern. Lowers to one
        strided copy per input (kCopy2D)."""
        shapes = [self.tensors[x].shape for x in xs]
        lead = shapes[0][:-1]
        assert all(s[:-1] == lead for s in shapes), \
            f"concat: leading dims differ: {shapes}"
        ctot = sum(s[-1] for s in shapes)
        return self._emit("concat", list(xs), (*lead, ctot), {}, name)

    def attention(self, qkv: str, heads: int, seq: int,
                  varlen: bool = False, pad_id: int = 0,
                  causal: bool = False,
                  name: Optional[str] = None) -> str:
        """varlen: mask keys beyond each sequence's valid length (derived
        from right-padded token ids; requires an i32 ids graph input).
        causal: decoder-style key > query masking (GPT-family)."""
        m, k3 = self.tensors[qkv].shape
        hid = k3 // 3
        return self._emit("attention", [qkv], (m, hid),
                          dict(heads=heads, seq=seq, head_dim=hid // heads,
                               varlen=varlen, pad_id=pad_id,
                               causal=causal),
                          name)

    # ------------------------------------------------------------ helpers
    def users(self, tensor: str) -> List[Node]:
        return [n for n in self.nodes if tensor in n.inputs]
