"""Planner: IR graph -> executable plan.

Passes (mirrors what TensorRT did internally for the reference, made
explicit and MI355X-shaped):
  1. fusion — conv2d+BN(+ReLU)(+residual add) into one implicit-GEMM kernel
     with a fused epilogue; gemm+bias(+ReLU/GeLU) likewise.
  2. legalization — pad input channels to C % 8 == 0 (the implicit-GEMM
     staging loads 16 B per lane), assert K % 64 == 0 for GEMM paths.
  3. weight prepacking — conv [Cout,Cin,KH,KW] -> [Cout][KH*KW*Cpad] fp16
     "bt" layout (K padded to 64); BN folded to per-channel scale/bias fp32.
  4. memory planning — liveness intervals + best-fit arena offsets
     (trtlab_amd.memory.ArenaPlanner): the activation-arena contract of the
     reference (workspace.cc:40-41 setDeviceMemory).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional, Tuple

import numpy as np

from trtlab_amd.engine.ir import Graph, Node
from trtlab_amd.memory import ArenaPlanner
from trtlab_amd.utils import round_up

# epilogue codes — keep in sync with csrc/kernels/gemm_common.h Epi
EPI_NONE = 0
EPI_BIAS = 1
EPI_BIAS_RELU = 2
EPI_BIAS_GELU = 3
EPI_SCALE_BIAS = 4
EPI_SCALE_BIAS_RELU = 5
EPI_SCALE_BIAS_ADD_RELU = 6
EPI_SCALE_BIAS_GELU = 7

# op kinds — keep in sync with csrc/runtime/runtime.h OpKind
K_CONV, K_GEMM, K_MAXPOOL, K_GAVGPOOL, K_SOFTMAX, K_LAYERNORM, \
    K_ADD_LAYERNORM, K_ELEMENTWISE, K_CHANNEL_PAD, K_ATTENTION = range(10)

DT_F16 = 0
DT_BF16 = 1
DT_I8 = 2
DT_F8 = 3  # OCP fp8 e4m3
DT_MX4 = 4  # OCP MXFP4 (e2m1 + e8m0 block scales) for GEMMs; rest fp16
DT_MX8 = 5  # OCP MXFP8 (e4m3 + e8m0 block scales) for GEMMs; rest fp16

K_QUANTIZE = 10
K_DEQUANT = 11
K_EMBEDDING = 12
K_AVGPOOL = 13
K_SEQLENS = 14  # token ids -> per-sequence valid length (varlen attention)
K_QUANT_MX4 = 15  # fp16 rows -> MXFP4 codes + e8m0 block scales
K_GEMM_MX4 = 16  # MXFP4 x MXFP4 scaled-MFMA GEMM (fp16 out + epilogue)
K_QUANT_MX8 = 17  # fp16 rows -> MXFP8 (e4m3) codes + e8m0 block scales
K_GEMM_MX8 = 18  # MXFP8 x MXFP8 scaled-MFMA GEMM (fp16 out + epilogue)
K_CLIP = 19  # out = min(max(x, mn), mx) — ONNX Clip with arbitrary bounds
K_TRANSPOSE2D = 20  # out[N][M] = in[M][N]^T (tiled LDS transpose kernel)
K_COPY2D = 21  # dst[m][coff + c] = src[m][c] — Concat lowering
K_RMSNORM = 22  # LLaMA norm: x / rms(x) * gamma
K_SILU_MUL = 23  # SwiGLU gate: silu(a) * b
K_ROPE = 24  # rotary embedding, in-place on qkv (arena-aliased output)
K_BTAIL = 25  # fused bottleneck tail: conv3x3+BN+ReLU -> 1x1+BN+res+ReLU
K_CONST = 26  # weight-blob constant -> arena tensor (one D2D copy)
K_VIEW = 27   # zero-copy reshape: output aliases the input's arena bytes
K_CHAFF = 28  # per-channel affine (+ReLU): standalone (pre-act) batchnorm


def _bf16_bits(arr: np.ndarray) -> np.ndarray:
    """fp16/fp32 -> bf16 bit patterns as int16 (numpy has no bf16 dtype;
    torch supplies the round-to-nearest-even conversion)."""
    import torch

    t = torch.from_numpy(np.ascontiguousarray(arr, np.float32))
    return t.to(torch.bfloat16).view(torch.int16).numpy()


@dataclass
class ExecOp:
    kind: int
    name: str
    inputs: List[str]            # tensor names (in, [in2/residual])
    output: str
    params: Dict[str, Any] = field(default_factory=dict)
    # weight payloads (numpy), packed into the blob by finalize()
    w: Optional[np.ndarray] = None        # fp16 bt-packed weights
    scale: Optional[np.ndarray] = None    # fp32 per-channel
    bias: Optional[np.ndarray] = None     # fp32 per-channel


@dataclass
class EnginePlan:
    name: str
    ops: List[Dict[str, Any]]            # dicts for _C.Engine
    exec_ops: List[ExecOp]               # named ops (CPU reference, tests)
    weights: np.ndarray                  # uint8 blob
    arena_bytes: int
    offsets: Dict[str, int]
    input_name: str
    input_off: int
    input_bytes: int
    input_shape: Tuple[int, ...]
    output_name: str
    output_off: int
    output_bytes: int
    output_shape: Tuple[int, ...]
    dtype: int = DT_F16
    shapes: Dict[str, Tuple[int, ...]] = field(default_factory=dict)
    input_dtype: str = "f16"
    # N-binding I/O (reference bindings.h:60-120): ordered binding dicts
    # {name, off, bytes, shape, dtype}; [0] is the primary binding the
    # legacy single-binding fields mirror.
    inputs: List[Dict[str, Any]] = field(default_factory=list)
    outputs: List[Dict[str, Any]] = field(default_factory=list)

    def __post_init__(self):
        if not self.inputs:
            self.inputs = [dict(name=self.input_name, off=self.input_off,
                                bytes=self.input_bytes,
                                shape=tuple(self.input_shape),
                                dtype=self.input_dtype)]
        if not self.outputs:
            self.outputs = [dict(name=self.output_name, off=self.output_off,
                                 bytes=self.output_bytes,
                                 shape=tuple(self.output_shape),
                                 dtype="bf16" if self.dtype == DT_BF16
                                 else "f16")]


class Planner:
    """dtype=DT_I8 lowers the conv stack to int8 (symmetric, per-channel
    weights, per-tensor activations calibrated on a synthetic forward);
    pools stay int8, the classifier head (gavgpool/gemm/softmax) stays fp16
    with quantize/dequant staging ops — BASELINE config 3."""

    def __init__(self, dtype: int = DT_F16, reuse: bool = True,
                 calib_sample=None, fork_join: bool = False,
                 btail_fusion: bool = True):
        self.dtype = dtype
        self.reuse = reuse  # False: disjoint arena slots (debugging)
        self.calib_sample = calib_sample  # int8 activation calibration input
        # fork_join: dual-stream downsample overlap inside the captured
        # graph. MEASURED ON MI355X (gpurun_out/check4): it REGRESSES
        # pipelined throughput 20.4k -> 13.1k inf/s rn50 b8 — with 3
        # contexts in flight the GPU is already saturated and the event
        # nodes + CU contention outweigh the overlap. Keep OFF for serving;
        # useful only for single-inflight latency experiments.
        self.fork_join = fork_join
        # bottleneck-tail fusion (fp16; measured +1.3x on the stage-1
        # pair — see the pass below); off switch for A/B benchmarks
        self.btail_fusion = btail_fusion

    # ------------------------------------------------------------- fusion
    def fuse(self, g: Graph) -> List[ExecOp]:
        nodes = g.nodes
        consumed: set = set()
        exec_ops: List[ExecOp] = []

        def single_user(t: str) -> Optional[Node]:
            us = g.users(t)
            return us[0] if len(us) == 1 else None

        for n in nodes:
            if n.name in consumed or n.kind == "input":
                continue
            if n.kind == "conv2d":
                exec_ops.append(self._fuse_conv(g, n, consumed, single_user))
            elif n.kind == "gemm":
                exec_ops.append(self._fuse_gemm(g, n, consumed, single_user))
            elif n.kind == "batchnorm":
                # standalone BN (producer is not a conv — DenseNet's
                # pre-activation blocks after concat): fold the stats
                # into a per-channel affine; absorb a following relu
                a = n.attrs
                sc = (a["gamma"] /
                      np.sqrt(a["var"] + a["eps"])).astype(np.float32)
                bi = (a["beta"] - a["mean"] * sc).astype(np.float32)
                out = n.output
                relu = 0
                r = single_user(out)
                if r is not None and r.kind == "relu":
                    consumed.add(r.name)
                    out = r.output
                    relu = 1
                op = ExecOp(K_CHAFF, n.name, [n.inputs[0]], out,
                            dict(relu=relu))
                op.scale, op.bias = sc, bi
                exec_ops.append(op)
            elif n.kind == "relu":
                exec_ops.append(ExecOp(K_ELEMENTWISE, n.name, [n.inputs[0]],
                                       n.output, dict(op=0)))
            elif n.kind == "gelu":
                exec_ops.append(ExecOp(K_ELEMENTWISE, n.name, [n.inputs[0]],
                                       n.output, dict(op=1)))
            elif n.kind == "add":
                exec_ops.append(ExecOp(K_ELEMENTWISE, n.name, list(n.inputs),
                                       n.output, dict(op=2)))
            elif n.kind == "maxpool":
                exec_ops.append(ExecOp(K_MAXPOOL, n.name, [n.inputs[0]],
                                       n.output, dict(n.attrs)))
            elif n.kind == "avgpool":
                exec_ops.append(ExecOp(K_AVGPOOL, n.name, [n.inputs[0]],
                                       n.output, dict(n.attrs)))
            elif n.kind == "gavgpool":
                exec_ops.append(ExecOp(K_GAVGPOOL, n.name, [n.inputs[0]],
                                       n.output, dict(n.attrs)))
            elif n.kind == "softmax":
                exec_ops.append(ExecOp(K_SOFTMAX, n.name, [n.inputs[0]],
                                       n.output, {}))
            elif n.kind == "layernorm":
                op = ExecOp(K_LAYERNORM, n.name, [n.inputs[0]], n.output,
                            dict(eps=n.attrs["eps"]))
                op.scale = n.attrs["gamma"].astype(np.float32)
                op.bias = n.attrs["beta"].astype(np.float32)
                exec_ops.append(op)
            elif n.kind == "add_layernorm":
                op = ExecOp(K_ADD_LAYERNORM, n.name, list(n.inputs), n.output,
                            dict(eps=n.attrs["eps"]))
                op.scale = n.attrs["gamma"].astype(np.float32)
                op.bias = n.attrs["beta"].astype(np.float32)
                exec_ops.append(op)
            elif n.kind == "embedding":
                # inputs = [ids] or [ids, segids] (segids -> in2_off)
                op = ExecOp(K_EMBEDDING, n.name, list(n.inputs), n.output,
                            dict(seq=n.attrs["seq"]))
                op.w = n.attrs["tok"].astype(np.float16)
                op.scale = n.attrs["pos"].astype(np.float16)
                op.bias = (None if n.attrs["seg"] is None
                           else n.attrs["seg"].astype(np.float16))
                exec_ops.append(op)
            elif n.kind == "attention":
                exec_ops.append(ExecOp(K_ATTENTION, n.name, [n.inputs[0]],
                                       n.output, dict(n.attrs)))
            elif n.kind == "rmsnorm":
                op = ExecOp(K_RMSNORM, n.name, [n.inputs[0]], n.output,
                            dict(eps=n.attrs["eps"]))
                op.scale = n.attrs["gamma"].astype(np.float32)
                exec_ops.append(op)
            elif n.kind == "silu_mul":
                exec_ops.append(ExecOp(K_SILU_MUL, n.name, list(n.inputs),
                                       n.output, {}))
            elif n.kind == "rope":
                exec_ops.append(ExecOp(K_ROPE, n.name, [n.inputs[0]],
                                       n.output, dict(n.attrs)))
            elif n.kind == "clip":
                exec_ops.append(ExecOp(K_CLIP, n.name, [n.inputs[0]],
                                       n.output,
                                       dict(mn=n.attrs["mn"],
                                            mx=n.attrs["mx"])))
            elif n.kind == "transpose2d":
                exec_ops.append(ExecOp(K_TRANSPOSE2D, n.name, [n.inputs[0]],
                                       n.output, {}))
            elif n.kind == "concat":
                # one strided copy per input into its column range
                coff = 0
                ctot = g.tensors[n.output].shape[-1]
                for ci, t in enumerate(n.inputs):
                    c = g.tensors[t].shape[-1]
                    exec_ops.append(ExecOp(
                        K_COPY2D, f"{n.name}_part{ci}", [t], n.output,
                        dict(coff=coff, C=c, ldd=ctot)))
                    coff += c
            elif n.kind == "view":
                # zero-copy reshape (row-major layouts agree); lowered to
                # an arena alias like rope's in-place output
                op = ExecOp(K_VIEW, n.name, [n.inputs[0]], n.output,
                            dict(shape=g.tensors[n.output].shape))
                exec_ops.append(op)
            elif n.kind == "constant":
                # device-resident constant (e.g. ViT position embeddings):
                # packed into the weight blob, copied D2D into its arena
                # slot each run (captured in the graph)
                op = ExecOp(K_CONST, n.name, [], n.output,
                            dict(shape=g.tensors[n.output].shape))
                op.w = n.attrs["value"].astype(np.float16)
                exec_ops.append(op)
            elif n.kind == "flatten":
                raise ValueError("flatten should be a view, not a node")
            else:
                raise ValueError(f"unknown node kind {n.kind}")
        return exec_ops

    def _fuse_conv(self, g: Graph, conv: Node, consumed: set, single_user):
        def_index = {n.output: i for i, n in enumerate(g.nodes)}
        conv_idx = def_index[conv.output]
        out = conv.output
        scale = bias = None
        residual = None
        epi = EPI_NONE
        # conv -> batchnorm?
        bn = single_user(out)
        if bn is not None and bn.kind == "batchnorm":
            a = bn.attrs
            s = (a["gamma"] / np.sqrt(a["var"] + a["eps"])).astype(np.float32)
            b = (a["beta"] - a["mean"] * s).astype(np.float32)
            scale, bias = s, b
            consumed.add(bn.name)
            out = bn.output
            epi = EPI_SCALE_BIAS
        # -> add(residual)?
        nxt = single_user(out)
        if nxt is not None and nxt.kind == "add" and scale is not None:
            other = nxt.inputs[0] if nxt.inputs[1] == out else nxt.inputs[1]
            rl = single_user(nxt.output)
            # the residual must already be computed when this conv runs
            if rl is not None and rl.kind == "relu" and \
                    def_index.get(other, 1 << 30) < conv_idx:
                residual = other
                consumed.add(nxt.name)
                consumed.add(rl.name)
                out = rl.output
                epi = EPI_SCALE_BIAS_ADD_RELU
        # -> relu?
        if epi in (EPI_NONE, EPI_SCALE_BIAS):
            r = single_user(out)
            if r is not None and r.kind == "relu":
                consumed.add(r.name)
                out = r.output
                epi = EPI_SCALE_BIAS_RELU if scale is not None else EPI_BIAS_RELU
                if scale is None:
                    cout = conv.attrs["weight"].shape[0]
                    bias = np.zeros(cout, np.float32)

        inputs = [conv.inputs[0]] + ([residual] if residual else [])
        op = ExecOp(K_CONV, conv.name, inputs, out,
                    dict(stride=conv.attrs["stride"],
                         padding=conv.attrs["padding"],
                         weight_shape=conv.attrs["weight"].shape, epi=epi))
        op.w = conv.attrs["weight"]  # packed in finalize
        op.scale, op.bias = scale, bias
        return op

    def _fuse_gemm(self, g: Graph, gm: Node, consumed: set, single_user):
        out = gm.output
        bias = gm.attrs.get("bias")
        epi = EPI_BIAS if bias is not None else EPI_NONE
        nxt = single_user(out)
        if nxt is not None and nxt.kind in ("relu", "gelu"):
            consumed.add(nxt.name)
            out = nxt.output
            if bias is None:
                bias = np.zeros(gm.attrs["weight"].shape[0], np.float32)
            epi = EPI_BIAS_RELU if nxt.kind == "relu" else EPI_BIAS_GELU
        op = ExecOp(K_GEMM, gm.name, [gm.inputs[0]], out,
                    dict(weight_shape=gm.attrs["weight"].shape, epi=epi))
        op.w = gm.attrs["weight"]
        op.bias = None if bias is None else bias.astype(np.float32)
        return op

    # ---------------------------------------------------------- MX lowering
    def _lower_mx(self, exec_ops, shapes, itemsize, fp4: bool):
        """Lower eligible GEMMs to OCP Microscaling (fp4-e2m1 or fp8-e4m3
        elements + e8m0 per-32-block scales) on the CDNA4 scaled MFMA:
        weights are block-quantized at build time; a quantize op converts
        the activation rows on device before each lowered GEMM. Epilogues
        (bias/GeLU) stay fused. Everything else stays fp16."""
        from trtlab_amd.engine.mx import quantize_mxfp4, quantize_mxfp8

        kmult = 256 if fp4 else 128  # one LDS tile of logical K
        quantize = quantize_mxfp4 if fp4 else quantize_mxfp8
        kq = K_QUANT_MX4 if fp4 else K_QUANT_MX8
        kg = K_GEMM_MX4 if fp4 else K_GEMM_MX8
        new_ops = []
        for op in exec_ops:
            m, k = (None, None)
            if op.kind == K_GEMM:
                m, k = shapes[op.inputs[0]]
            if op.kind != K_GEMM or k % kmult != 0 or \
                    op.params["epi"] not in (EPI_NONE, EPI_BIAS,
                                             EPI_BIAS_RELU, EPI_BIAS_GELU):
                new_ops.append(op)
                continue
            nout = op.params["weight_shape"][0]
            wq, wsc = quantize(op.w.astype(np.float32))
            codes_t = op.name + "_mxq"
            scales_t = op.name + "_mxs"
            shapes[codes_t] = (m, k // 2 if fp4 else k)
            shapes[scales_t] = (m, k // 32)
            itemsize[codes_t] = 1
            itemsize[scales_t] = 1
            q = ExecOp(kq, op.name + "_quant", [op.inputs[0]],
                       codes_t, dict(M=m, K=k, q_out=scales_t))
            g = ExecOp(kg, op.name, [codes_t, scales_t], op.output,
                       dict(epi=op.params["epi"], M=m, N=nout, K=k,
                            mx=True))
            g.w = wq                     # packed element codes
            g.bias = op.bias             # fp32 epilogue bias
            g.params["mx_wscales"] = wsc  # e8m0 [N, K/32] -> w2_off
            new_ops.append(q)
            new_ops.append(g)
        # ---- producer fusion: a quant whose input row is produced by a
        # layernorm/add_layernorm is folded INTO that LN (the kernel emits
        # the MX codes + e8m0 scales alongside its fp16 output), dropping
        # the standalone quantize launch (round-2 MX lever).
        producers = {}
        for op in new_ops:
            producers[op.output] = op
        fused_ops = []
        mode = 4 if fp4 else 8
        for op in new_ops:
            if op.kind == kq:
                prod = producers.get(op.inputs[0])
                if prod is not None and prod.kind in (K_LAYERNORM,
                                                      K_ADD_LAYERNORM) \
                        and "mx_out" not in prod.params:
                    prod.params["mx_out"] = op.output
                    prod.params["mx_scales"] = op.params["q_out"]
                    prod.params["mx_mode"] = mode
                    producers[op.output] = prod
                    continue  # quant op absorbed
            fused_ops.append(op)
        exec_ops[:] = fused_ops

    # --------------------------------------------------------------- plan
    def compile(self, g: Graph) -> EnginePlan:
        exec_ops = self.fuse(g)
        shapes: Dict[str, Tuple[int, ...]] = {
            t: spec.shape for t, spec in g.tensors.items()
        }

        # ---- legalization: pad input channels for conv staging ----
        input_name = g.input_name
        assert input_name is not None
        in_shape = shapes[input_name]
        padded_input = input_name
        pad_mult = 16 if self.dtype in (DT_I8, DT_F8) else 8
        if len(in_shape) == 4 and in_shape[3] % pad_mult != 0:
            cpad = round_up(in_shape[3], pad_mult)
            padded_input = input_name + "_padded"
            shapes[padded_input] = (*in_shape[:3], cpad)
            m = in_shape[0] * in_shape[1] * in_shape[2]
            pad_op = ExecOp(K_CHANNEL_PAD, padded_input, [input_name],
                            padded_input,
                            dict(M=m, Cin=in_shape[3], Cpad=cpad))
            exec_ops.insert(0, pad_op)
            for op in exec_ops[1:]:
                op.inputs = [padded_input if t == input_name else t
                             for t in op.inputs]

        # ---- weight prepacking (+ int8 lowering) ----
        itemsize: Dict[str, int] = {
            t: {"f16": 2, "bf16": 2, "f32": 4, "i32": 4, "i8": 1}[spec.dtype]
            for t, spec in g.tensors.items()
        }

        # ---- variable-length attention: derive per-sequence lengths ----
        # one seqlens op per forward (ids -> [B] i32 in the arena); every
        # varlen attention op reads it as a second input (in2_off).
        att_varlen = [op for op in exec_ops
                      if op.kind == K_ATTENTION and op.params.get("varlen")]
        if att_varlen:
            # lengths come from a dedicated attention_mask binding when the
            # graph has one (count of non-zero entries == pad_id 0 scan),
            # else from the token-id input vs pad_id
            if "attention_mask" in g.input_names:
                lens_src = "attention_mask"
                pad_id = 0
            else:
                lens_src = input_name
                pad_id = att_varlen[0].params.get("pad_id", 0)
            if g.tensors[lens_src].dtype != "i32":
                raise ValueError(
                    "varlen attention requires an i32 token-id or "
                    "attention_mask graph input")
            seq = att_varlen[0].params["seq"]
            bsz = shapes[lens_src][0] // seq
            lens_name = "_seqlens"
            shapes[lens_name] = (bsz,)
            itemsize[lens_name] = 4
            exec_ops.insert(0, ExecOp(
                K_SEQLENS, lens_name, [lens_src], lens_name,
                dict(B=bsz, S=seq, pad_id=pad_id)))
            for op in att_varlen:
                op.inputs.append(lens_name)
        if self.dtype in (DT_MX4, DT_MX8):
            self._lower_mx(exec_ops, shapes, itemsize,
                           fp4=self.dtype == DT_MX4)
        if self.dtype in (DT_I8, DT_F8):
            from trtlab_amd.engine.quantize import lower_int8

            lower_int8(g, exec_ops, shapes, itemsize, input_name,
                       padded_input, self.calib_sample,
                       fmt="f8" if self.dtype == DT_F8 else "i8")
        # ---- bottleneck-tail fusion (fp16 only): a 3x3/s1/p1 conv with
        # BN+ReLU whose 64-wide output feeds exactly one 1x1 conv with
        # BN+residual+ReLU collapses into ONE kernel — the intermediate
        # tensor lives in LDS (csrc bottleneck_tail_kernel). MEASURED
        # (tools/dbg_btail, MI355X b8): stage-1 pair 30.0 -> 22.8 us
        # (1.31x); the 128-wide stage-2 variant LOSES from grid
        # starvation (98-196 WGs), so only Cm == 64 fuses.
        if self.dtype == DT_F16 and self.btail_fusion:
            cons_count: Dict[str, int] = {}
            for o in exec_ops:
                for tt in o.inputs:
                    cons_count[tt] = cons_count.get(tt, 0) + 1
            pinned_outs = set(g.output_names) | {g.output_name}
            i = 0
            while i < len(exec_ops):
                a = exec_ops[i]
                merged = False
                if (a.kind == K_CONV and a.params.get("int8") is None and
                        a.params.get("epi") == EPI_SCALE_BIAS_RELU):
                    ws = a.params["weight_shape"]
                    if (ws[2] == 3 and ws[3] == 3 and
                            ws[0] == 64 and ws[1] == 64 and
                            a.params["stride"] == 1 and
                            a.params["padding"] == 1 and
                            cons_count.get(a.output, 0) == 1 and
                            a.output not in pinned_outs):
                        j = next((k for k in range(i + 1, len(exec_ops))
                                  if a.output in exec_ops[k].inputs), None)
                        if j is not None:
                            b = exec_ops[j]
                            wb = b.params.get("weight_shape")
                            if (b.kind == K_CONV and wb is not None and
                                    b.params.get("int8") is None and
                                    b.params.get("epi") ==
                                    EPI_SCALE_BIAS_ADD_RELU and
                                    wb[2] == 1 and wb[3] == 1 and
                                    wb[1] == 64 and
                                    b.params["stride"] == 1 and
                                    b.inputs[0] == a.output and
                                    len(b.inputs) > 1):
                                co = int(wb[0])
                                f = ExecOp(
                                    K_BTAIL, b.name,
                                    [a.inputs[0], b.inputs[1]], b.output,
                                    dict(Cm=64, Co=co))
                                f.w = np.ascontiguousarray(
                                    a.w.transpose(0, 2, 3, 1)
                                    .reshape(64, 576), np.float16)
                                f.params["w2"] = np.ascontiguousarray(
                                    b.w.reshape(co, 64), np.float16)
                                f.scale = np.concatenate(
                                    [a.scale, b.scale]).astype(np.float32)
                                f.bias = np.concatenate(
                                    [a.bias, b.bias]).astype(np.float32)
                                exec_ops[j] = f
                                del exec_ops[i]
                                merged = True
                if not merged:
                    i += 1

        for op in exec_ops:
            if op.kind == K_CONV and op.params.get("int8") is None:
                w = op.w  # [Cout, Cin, KH, KW] fp32
                cout, cin, kh, kw = w.shape
                cpad = round_up(cin, 8)
                whwc = np.transpose(w, (0, 2, 3, 1))  # [Cout, KH, KW, Cin]
                if cpad != cin:
                    whwc = np.pad(whwc, ((0, 0), (0, 0), (0, 0), (0, cpad - cin)))
                k = kh * kw * cpad
                kp = round_up(k, 64)
                flat = whwc.reshape(cout, k)
                if kp != k:
                    flat = np.pad(flat, ((0, 0), (0, kp - k)))
                op.w = np.ascontiguousarray(flat, np.float16)
                op.params["C"] = cpad
                op.params["Kp"] = kp
            elif op.kind == K_GEMM and op.params.get("int8") is None:
                w = op.w  # [Nout, K] fp32
                nout, k = w.shape
                if k % 64 != 0:
                    raise ValueError(f"gemm {op.name}: K={k} must be %64")
                op.w = np.ascontiguousarray(w, np.float16)

        # ---- weight blob ----
        blob = bytearray()

        def pack(arr: Optional[np.ndarray]) -> int:
            if arr is None:
                return -1
            off = round_up(len(blob), 256)
            blob.extend(b"\0" * (off - len(blob)))
            blob.extend(np.ascontiguousarray(arr).tobytes())
            return off

        def pack_half(arr: Optional[np.ndarray]) -> int:
            # compute-dtype tensor slot: fp16 numeric in exec_ops (so the
            # CPU reference executor and autotune stay numeric), re-encoded
            # to bf16 bit patterns in the blob for DT_BF16 plans
            if arr is not None and self.dtype == DT_BF16 and \
                    arr.dtype in (np.float16, np.float32):
                return pack(_bf16_bits(arr))
            return pack(arr)

        w_offs: Dict[str, Tuple[int, int, int]] = {}
        w2_offs: Dict[str, int] = {}
        for op in exec_ops:
            if op.kind == K_EMBEDDING:  # pos/seg tables are compute-dtype
                w_offs[op.name] = (pack_half(op.w), pack_half(op.scale),
                                   pack_half(op.bias))
            else:  # scale/bias are fp32 epilogue params, w is compute-dtype
                w_offs[op.name] = (pack_half(op.w), pack(op.scale),
                                   pack(op.bias))
            if op.params.get("mx_wscales") is not None:
                w2_offs[op.name] = pack(op.params["mx_wscales"])
            elif op.params.get("w2") is not None:
                w2_offs[op.name] = pack_half(op.params["w2"])

        # ---- fork/join dual-stream schedule (downsample overlap) ----
        # A conv whose output's SOLE use is the residual (in2) input of a
        # later conv runs on the context's side stream, overlapping the
        # main bottleneck chain (the ResNet downsample pattern). Pairs are
        # strictly sequential; the executor turns them into hipEvent
        # dependencies inside the captured graph (executor.cpp fork/join).
        consumers_of: Dict[str, List[int]] = {}
        for i, op in enumerate(exec_ops):
            for t in op.inputs:
                consumers_of.setdefault(t, []).append(i)
        fork_pairs: List[Tuple[int, int]] = []
        last_join = -1
        for i, op in enumerate(exec_ops):
            if not self.fork_join:
                break
            if op.kind != K_CONV or i <= last_join:
                continue
            cons = consumers_of.get(op.output, [])
            if len(cons) != 1:
                continue
            j = cons[0]
            cj = exec_ops[j]
            if j <= i + 1 or cj.kind not in (K_CONV, K_BTAIL):
                continue
            if len(cj.inputs) < 2 or cj.inputs[1] != op.output:
                continue
            op.params["fork"] = 1
            cj.params["join"] = 1
            fork_pairs.append((i, j))
            last_join = j

        # ---- liveness + arena offsets ----
        tensors_used: Dict[str, Tuple[int, int]] = {}

        def touch(t: str, i: int):
            if t in tensors_used:
                s, e = tensors_used[t]
                tensors_used[t] = (min(s, i), max(e, i))
            else:
                tensors_used[t] = (i, i)

        input_names = list(g.input_names) or [input_name]
        for t in input_names:
            touch(t, 0)  # all input bindings live from the start (H2D)
        for i, op in enumerate(exec_ops):
            for t in op.inputs:
                touch(t, i)
            touch(op.output, i)
            if "q_out" in op.params:  # fused fp8 second output
                touch(op.params["q_out"], i)
            if "mx_out" in op.params:  # producer-fused MX outputs
                touch(op.params["mx_out"], i)
                touch(op.params["mx_scales"], i)
        output_name = exec_ops[-1].output
        # N output bindings: the final op's output is the primary binding
        # [0]; tensors pinned by g.mark_output follow. All live to the end
        # (each gets a D2H copy).
        output_names = [output_name] + [t for t in g.output_names
                                        if t != output_name]
        for t in output_names:
            s, e = tensors_used[t]
            tensors_used[t] = (s, len(exec_ops))
        # forked ops run concurrently with ops (i, j): their INPUT regions
        # must stay live until the join so no intermediate output aliases
        # memory the side stream is still reading
        for fi, fj in fork_pairs:
            for t in exec_ops[fi].inputs:
                if t in tensors_used:
                    s, e = tensors_used[t]
                    tensors_used[t] = (s, max(e, fj))

        def nbytes_of(t: str) -> int:
            n = 1
            for d in shapes[t]:
                n *= d
            return n * itemsize.get(t, 2)

        # rope outputs alias their inputs (in-place rotation): merge the
        # two tensors' live intervals onto the INPUT and copy its offset
        # to the output after planning
        rope_alias: Dict[str, str] = {}
        for op in exec_ops:
            if op.kind in (K_ROPE, K_VIEW):
                src, dst = op.inputs[0], op.output
                # follow chains (rope of rope never happens, but be safe)
                src = rope_alias.get(src, src)
                rope_alias[dst] = src
                s0a, e0a = tensors_used[src]
                s0b, e0b = tensors_used[dst]
                tensors_used[src] = (min(s0a, s0b), max(e0a, e0b))
                del tensors_used[dst]

        arena = ArenaPlanner()
        for t, (s0, e0) in tensors_used.items():
            if self.reuse:
                arena.add(t, nbytes_of(t), s0, e0)
            else:
                arena.add(t, nbytes_of(t), 0, len(exec_ops))
        offsets, arena_bytes = arena.plan()
        for dst, src in rope_alias.items():
            offsets[dst] = offsets[src]

        # ---- emit op dicts ----
        op_dicts: List[Dict[str, Any]] = []
        for op in exec_ops:
            w_off, s_off, b_off = w_offs[op.name]
            op_dtype = op.params.get(
                "dtype",
                DT_F16 if self.dtype in (DT_I8, DT_F8, DT_MX4, DT_MX8)
                else self.dtype)
            d: Dict[str, Any] = dict(dtype=op_dtype, w_off=w_off,
                                     scale_off=s_off, bias_off=b_off,
                                     in_off=(offsets[op.inputs[0]]
                                             if op.inputs else -1),
                                     out_off=offsets[op.output],
                                     fork=op.params.get("fork", 0),
                                     join=op.params.get("join", 0))
            if len(op.inputs) > 1:
                d["in2_off"] = offsets[op.inputs[1]]
            if op.kind == K_CONV:
                ish = shapes[op.inputs[0]]
                osh = shapes[op.output]
                d.update(kind=K_CONV, epi=op.params["epi"], Nb=ish[0],
                         H=ish[1], W=ish[2], C=op.params["C"],
                         Cout=osh[3], KH=op.params["weight_shape"][2],
                         KW=op.params["weight_shape"][3],
                         sh=op.params["stride"], sw=op.params["stride"],
                         ph=op.params["padding"], pw=op.params["padding"],
                         res_scale=op.params.get("res_scale", 1.0))
            elif op.kind == K_CHAFF:
                sh = shapes[op.inputs[0]]
                m = 1
                for dd in sh[:-1]:
                    m *= dd
                d.update(kind=K_CHAFF, epi=op.params["relu"], n_elems=m,
                         C=sh[-1])
            elif op.kind == K_VIEW:
                d.update(kind=K_VIEW)
            elif op.kind == K_CONST:
                d.update(kind=K_CONST, n_elems=nbytes_of(op.output))
            elif op.kind == K_BTAIL:
                ish = shapes[op.inputs[0]]
                d.update(kind=K_BTAIL, Nb=ish[0], H=ish[1], W=ish[2],
                         C=op.params["Cm"], Cout=op.params["Co"],
                         w2_off=w2_offs[op.name])
            elif op.kind == K_GEMM:
                m, k = shapes[op.inputs[0]]
                nout = shapes[op.output][1]
                d.update(kind=K_GEMM, epi=op.params["epi"], M=m, N=nout, K=k,
                         q_scale=op.params.get("out_scale", 1.0))
            elif op.kind in (K_MAXPOOL, K_AVGPOOL):
                ish = shapes[op.inputs[0]]
                d.update(kind=op.kind, Nb=ish[0], H=ish[1], W=ish[2],
                         C=ish[3], KH=op.params["kernel"],
                         KW=op.params["kernel"], sh=op.params["stride"],
                         sw=op.params["stride"], ph=op.params["padding"],
                         pw=op.params["padding"])
            elif op.kind == K_GAVGPOOL:
                ish = shapes[op.inputs[0]]
                d.update(kind=K_GAVGPOOL, Nb=ish[0], HW=ish[1] * ish[2],
                         C=ish[3])
            elif op.kind == K_SOFTMAX:
                m, ncol = shapes[op.inputs[0]]
                d.update(kind=K_SOFTMAX, M=m, N=ncol)
            elif op.kind in (K_LAYERNORM, K_ADD_LAYERNORM):
                m, ncol = shapes[op.inputs[0]]
                d.update(kind=op.kind, M=m, N=ncol, eps=op.params["eps"])
                if "mx_out" in op.params:
                    # producer-fused MX: epi = mode (4/8), out2 = codes,
                    # out3 = e8m0 block scales
                    d.update(epi=op.params["mx_mode"],
                             out2_off=offsets[op.params["mx_out"]],
                             out3_off=offsets[op.params["mx_scales"]])
                else:
                    d.update(out2_off=offsets.get(op.params.get("q_out"),
                                                  -1),
                             q_scale=op.params.get("q_scale", 0.0))
            elif op.kind == K_ELEMENTWISE:
                n = 1
                for s_ in shapes[op.output]:
                    n *= s_
                d.update(kind=K_ELEMENTWISE, epi=op.params["op"], n_elems=n)
            elif op.kind == K_CHANNEL_PAD:
                d.update(kind=K_CHANNEL_PAD, n_elems=op.params["M"],
                         C=op.params["Cin"], Cout=op.params["Cpad"])
            elif op.kind == K_EMBEDDING:
                m, h = shapes[op.output]
                d.update(kind=K_EMBEDDING, M=m, S=op.params["seq"], N=h)
            elif op.kind == K_SEQLENS:
                d.update(kind=K_SEQLENS, B=op.params["B"], S=op.params["S"],
                         epi=op.params["pad_id"])
            elif op.kind in (K_QUANT_MX4, K_QUANT_MX8):
                d.update(kind=op.kind, M=op.params["M"],
                         K=op.params["K"],
                         out2_off=offsets[op.params["q_out"]])
            elif op.kind in (K_GEMM_MX4, K_GEMM_MX8):
                d.update(kind=op.kind, epi=op.params["epi"],
                         M=op.params["M"], N=op.params["N"],
                         K=op.params["K"], w2_off=w2_offs[op.name])
            elif op.kind in (K_QUANTIZE, K_DEQUANT):
                n = 1
                for s_ in shapes[op.output]:
                    n *= s_
                d.update(kind=op.kind, n_elems=n,
                         q_scale=op.params["q_scale"],
                         epi=1 if op.params.get("fmt") == "f8" else 0)
            elif op.kind == K_ATTENTION:
                m, hid = shapes[op.output]
                heads = op.params["heads"]
                seq = op.params["seq"]
                hd = op.params["head_dim"]
                d.update(kind=K_ATTENTION, B=m // seq, S=seq, NH=heads, HD=hd,
                         att_scale=1.0 / float(np.sqrt(hd)),
                         epi=op.params.get("out_dtype", 0),
                         q_scale=op.params.get("q_scale", 0.0),
                         causal=1 if op.params.get("causal") else 0)
            elif op.kind == K_RMSNORM:
                m, ncol = shapes[op.inputs[0]]
                d.update(kind=K_RMSNORM, M=m, N=ncol, eps=op.params["eps"])
            elif op.kind == K_SILU_MUL:
                n_ = 1
                for s_ in shapes[op.output]:
                    n_ *= s_
                d.update(kind=K_SILU_MUL, n_elems=n_)
            elif op.kind == K_ROPE:
                m, n3 = shapes[op.inputs[0]]
                heads = op.params["heads"]
                hd = n3 // (3 * heads)
                # eps carries theta; the op runs IN PLACE on the aliased
                # arena buffer (out_off == in_off by construction below)
                d.update(kind=K_ROPE, M=m, S=op.params["seq"], NH=heads,
                         HD=hd, eps=op.params["theta"])
            elif op.kind == K_CLIP:
                n = 1
                for s_ in shapes[op.output]:
                    n *= s_
                # res_scale/q_scale carry the clip bounds (see executor.cpp)
                d.update(kind=K_CLIP, n_elems=n, res_scale=op.params["mn"],
                         q_scale=op.params["mx"])
            elif op.kind == K_TRANSPOSE2D:
                m, ncol = shapes[op.inputs[0]]
                d.update(kind=K_TRANSPOSE2D, M=m, N=ncol)
            elif op.kind == K_COPY2D:
                rows = 1
                for s_ in shapes[op.inputs[0]][:-1]:
                    rows *= s_
                d.update(kind=K_COPY2D, M=rows, C=op.params["C"],
                         Cout=op.params["ldd"], epi=op.params["coff"])
            else:
                raise ValueError(f"bad exec op kind {op.kind}")
            op_dicts.append(d)

        # bf16 plans read float bindings as bf16 bits (the host side
        # converts); integer inputs (BERT token ids) stay as declared
        def bind_dtype(t: str) -> str:
            dt = g.tensors[t].dtype if t in g.tensors else "f16"
            return "bf16" if self.dtype == DT_BF16 and dt == "f16" else dt

        def binding(t: str) -> Dict[str, Any]:
            return dict(name=t, off=offsets[t], bytes=nbytes_of(t),
                        shape=tuple(shapes[t]), dtype=bind_dtype(t))

        return EnginePlan(
            name=g.name,
            ops=op_dicts,
            exec_ops=exec_ops,
            weights=np.frombuffer(bytes(blob), dtype=np.uint8).copy(),
            arena_bytes=arena_bytes,
            offsets=offsets,
            input_name=input_name,
            input_off=offsets[input_name],
            input_bytes=nbytes_of(input_name),
            input_shape=shapes[input_name],
            output_name=output_name,
            output_off=offsets[output_name],
            output_bytes=nbytes_of(output_name),
            output_shape=shapes[output_name],
            dtype=self.dtype,
            shapes=dict(shapes),
            input_dtype=bind_dtype(input_name),
            inputs=[binding(t) for t in input_names],
            outputs=[binding(t) for t in output_names],
        )
