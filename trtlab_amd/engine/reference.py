"""CPU reference executor: runs an EnginePlan's exec_ops with torch fp32.

Role: (a) validates the planner on CPU-only CI, (b) provides the numerics
reference the GPU engine is compared against in tests (the role ONNX test
vectors play in the reference's examples/ONNX/resnet50/run_onnx_tests.py).
Uses the SAME prepacked weights as the GPU path so the comparison isolates
kernel numerics.
"""
from __future__ import annotations

from typing import Dict

import numpy as np
import torch
import torch.nn.functional as F

from trtlab_amd.engine.planner import (
    K_CHAFF, K_CLIP, K_COPY2D, K_TRANSPOSE2D, K_RMSNORM, K_SILU_MUL, K_ROPE, K_VIEW,
    EnginePlan, K_ADD_LAYERNORM, K_ATTENTION, K_AVGPOOL, K_BTAIL, K_CHANNEL_PAD, K_CONST, K_CONV,
    K_DEQUANT, K_ELEMENTWISE, K_GAVGPOOL, K_GEMM, K_LAYERNORM, K_MAXPOOL,
    K_EMBEDDING, K_GEMM_MX4, K_GEMM_MX8, K_QUANT_MX4, K_QUANT_MX8,
    K_QUANTIZE, K_SEQLENS, K_SOFTMAX,
    EPI_BIAS, EPI_BIAS_GELU, EPI_BIAS_RELU, EPI_NONE, EPI_SCALE_BIAS,
    EPI_SCALE_BIAS_ADD_RELU, EPI_SCALE_BIAS_GELU, EPI_SCALE_BIAS_RELU)


def _fp8_round(x: torch.Tensor) -> torch.Tensor:
    """Round to the OCP e4m3 grid (saturating), staying in fp32."""
    return torch.clamp(x, -448, 448).to(torch.float8_e4m3fn).float()


def _epi(acc: torch.Tensor, epi: int, scale, bias, res) -> torch.Tensor:
    v = acc
    if epi in (EPI_SCALE_BIAS, EPI_SCALE_BIAS_RELU, EPI_SCALE_BIAS_ADD_RELU,
               EPI_SCALE_BIAS_GELU):
        v = v * scale
    if epi != EPI_NONE:
        v = v + bias
    if epi == EPI_SCALE_BIAS_ADD_RELU:
        v = v + res
    if epi in (EPI_BIAS_RELU, EPI_SCALE_BIAS_RELU, EPI_SCALE_BIAS_ADD_RELU):
        v = F.relu(v)
    if epi in (EPI_BIAS_GELU, EPI_SCALE_BIAS_GELU):
        v = F.gelu(v, approximate="tanh")
    return v


def run_reference(plan: EnginePlan, input_nhwc, return_all: bool = False):
    """Run the plan on CPU (fp32). input is the RAW (unpadded) primary
    input, or a {name: array} dict for multi-binding models. With
    return_all=True, returns the dict of ALL tensors (debugging)."""
    if isinstance(input_nhwc, dict):
        t: Dict[str, torch.Tensor] = {
            k: torch.from_numpy(np.ascontiguousarray(v)).float()
            for k, v in input_nhwc.items()
        }
    else:
        t = {plan.input_name:
             torch.from_numpy(np.ascontiguousarray(input_nhwc)).float()}
    # shapes registry from the planner's op dicts + exec op metadata
    for op, d in zip(plan.exec_ops, plan.ops):
        x = t[op.inputs[0]] if op.inputs else None
        if op.kind == K_CHANNEL_PAD:
            cin, cpad = d["C"], d["Cout"]
            flat = x.reshape(-1, cin)
            t[op.output] = F.pad(flat, (0, cpad - cin)).reshape(
                *x.shape[:-1], cpad)
        elif op.kind == K_CONV:
            nb, h, w, c = x.shape
            kh, kw = d["KH"], d["KW"]
            cout = d["Cout"]
            if d["dtype"] == 3:  # fp8 weights stored as uint8 codes
                wt = torch.from_numpy(op.w).view(torch.float8_e4m3fn).float()
            else:
                wt = torch.from_numpy(op.w.astype(np.float32))  # [Cout, Kp]
            k = kh * kw * c
            wt = wt[:, :k].reshape(cout, kh, kw, c).permute(0, 3, 1, 2)
            xc = x.permute(0, 3, 1, 2)
            acc = F.conv2d(xc, wt, stride=d["sh"], padding=d["ph"])
            acc = acc.permute(0, 2, 3, 1)  # NHWC
            scale = bias = res = None
            if op.scale is not None:
                scale = torch.from_numpy(op.scale)
            if op.bias is not None:
                bias = torch.from_numpy(op.bias)
            if len(op.inputs) > 1:
                res = t[op.inputs[1]]
                if d.get("res_scale", 1.0) != 1.0:
                    res = res * d["res_scale"]
            y = _epi(acc, d["epi"], scale, bias, res)
            if d["dtype"] == 2:  # int8: emulate the requantized store
                y = torch.clamp(torch.round(y), -127, 127)
            elif d["dtype"] == 3:  # fp8: emulate the e4m3 output store
                y = _fp8_round(y)
            t[op.output] = y
        elif op.kind == K_CHAFF:
            y = x * torch.from_numpy(op.scale) + torch.from_numpy(op.bias)
            t[op.output] = torch.relu(y) if d["epi"] else y
        elif op.kind == K_VIEW:
            t[op.output] = x.reshape(op.params["shape"])
        elif op.kind == K_CONST:
            t[op.output] = torch.from_numpy(
                op.w.astype(np.float32)).reshape(op.params["shape"])
        elif op.kind == K_BTAIL:
            # fused bottleneck tail: conv3x3+BN+ReLU then 1x1+BN+res+ReLU
            # (weights pre-packed: w = [64, 576] bt-flat, w2 = [Co, 64];
            # scale/bias = [s1 | s2] / [b1 | b2])
            nb, h, w, c = x.shape
            wt = torch.from_numpy(op.w.astype(np.float32)) \
                .reshape(c, 3, 3, c).permute(0, 3, 1, 2)
            mid = F.conv2d(x.permute(0, 3, 1, 2), wt, stride=1,
                           padding=1).permute(0, 2, 3, 1)
            sc = torch.from_numpy(op.scale)
            bi = torch.from_numpy(op.bias)
            mid = torch.relu(mid * sc[:c] + bi[:c])
            w2 = torch.from_numpy(
                op.params["w2"].astype(np.float32))  # [Co, c]
            co = d["Cout"]
            acc = mid.reshape(-1, c) @ w2.t()
            res = t[op.inputs[1]].reshape(-1, co)
            y = torch.relu(acc * sc[c:] + bi[c:] + res)
            t[op.output] = y.reshape(nb, h, w, co)
        elif op.kind == K_EMBEDDING:
            ids = x.long()
            tok = torch.from_numpy(op.w.astype(np.float32))
            pos = torch.from_numpy(op.scale.astype(np.float32))
            m = ids.shape[0]
            y = tok[ids] + pos[torch.arange(m) % d["S"]]
            if op.bias is not None:
                seg = torch.from_numpy(op.bias.astype(np.float32))
                if len(op.inputs) > 1:  # segids binding (token_type_ids)
                    y = y + seg[t[op.inputs[1]].long()]
                else:
                    y = y + seg[0]
            t[op.output] = y
        elif op.kind == K_QUANTIZE:
            if d.get("epi") == 1:  # fp8 e4m3 (continuous grid, sat 448)
                t[op.output] = _fp8_round(x / d["q_scale"])
            else:  # int8 codes
                t[op.output] = torch.clamp(torch.round(x / d["q_scale"]),
                                           -127, 127)
        elif op.kind == K_DEQUANT:
            t[op.output] = x * d["q_scale"]
        elif op.kind == K_GEMM:
            if d["dtype"] in (3, 4):  # fp8 compute (3: fp8 out, 4: fp16 out)
                wt = torch.from_numpy(op.w).view(torch.float8_e4m3fn).float()
                acc = x @ wt.t()
                scale = torch.from_numpy(op.scale)
                bias = torch.from_numpy(op.bias)
                y = _epi(acc, d["epi"], scale, bias, None)
                if d["dtype"] == 3:
                    y = _fp8_round(y * d.get("q_scale", 1.0))
                t[op.output] = y
            else:
                wt = torch.from_numpy(op.w.astype(np.float32))  # [N, K]
                acc = x @ wt.t()
                bias = (torch.from_numpy(op.bias)
                        if op.bias is not None else None)
                t[op.output] = _epi(acc, d["epi"], None, bias, None)
        elif op.kind == K_MAXPOOL:
            xc = x.permute(0, 3, 1, 2)
            y = F.max_pool2d(xc, d["KH"], stride=d["sh"], padding=d["ph"])
            t[op.output] = y.permute(0, 2, 3, 1)
        elif op.kind == K_AVGPOOL:
            xc = x.permute(0, 3, 1, 2)
            y = F.avg_pool2d(xc, d["KH"], stride=d["sh"], padding=d["ph"],
                             count_include_pad=False)
            t[op.output] = y.permute(0, 2, 3, 1)
        elif op.kind == K_GAVGPOOL:
            nb, h, w, c = x.shape
            t[op.output] = x.reshape(nb, h * w, c).mean(dim=1)
        elif op.kind == K_SOFTMAX:
            t[op.output] = F.softmax(x, dim=-1)
        elif op.kind in (K_LAYERNORM, K_ADD_LAYERNORM):
            src = x if op.kind == K_LAYERNORM else x + t[op.inputs[1]]
            g_ = torch.from_numpy(op.scale)
            b_ = torch.from_numpy(op.bias)
            y = F.layer_norm(src, (src.shape[-1],), g_, b_, d["eps"])
            t[op.output] = y
            if op.params.get("q_out"):
                t[op.params["q_out"]] = _fp8_round(y / d["q_scale"])
            if op.params.get("mx_out"):  # producer-fused MX quantization
                from trtlab_amd.engine.mx import (quantize_mxfp4,
                                                  quantize_mxfp8)

                qf = (quantize_mxfp4 if op.params["mx_mode"] == 4
                      else quantize_mxfp8)
                codes, scales = qf(y.numpy().astype(np.float32))
                t[op.params["mx_out"]] = torch.from_numpy(codes)
                t[op.params["mx_scales"]] = torch.from_numpy(scales)
        elif op.kind == K_ELEMENTWISE:
            code = d["epi"]
            if code == 0:
                t[op.output] = F.relu(x)
            elif code == 1:
                t[op.output] = F.gelu(x, approximate="tanh")
            elif code == 2:
                t[op.output] = x + t[op.inputs[1]]
            elif code == 3:
                t[op.output] = F.relu(x + t[op.inputs[1]])
        elif op.kind in (K_QUANT_MX4, K_QUANT_MX8):
            from trtlab_amd.engine.mx import quantize_mxfp4, quantize_mxfp8

            qf = quantize_mxfp4 if op.kind == K_QUANT_MX4 else quantize_mxfp8
            codes, scales = qf(x.numpy().astype(np.float32))
            t[op.output] = torch.from_numpy(codes)
            t[op.params["q_out"]] = torch.from_numpy(scales)
        elif op.kind in (K_GEMM_MX4, K_GEMM_MX8):
            from trtlab_amd.engine.mx import (dequantize_mxfp4,
                                              dequantize_mxfp8)

            df = (dequantize_mxfp4 if op.kind == K_GEMM_MX4
                  else dequantize_mxfp8)
            a = df(x.numpy(), t[op.inputs[1]].numpy())
            w = df(op.w, op.params["mx_wscales"])
            acc = torch.from_numpy(a) @ torch.from_numpy(w).T
            bias = (torch.from_numpy(op.bias) if op.bias is not None
                    else None)
            y = _epi(acc, d["epi"], None, bias, None)
            t[op.output] = y
        elif op.kind == K_SEQLENS:
            bsz, seq = d["B"], d["S"]
            lens = (x.reshape(bsz, seq).long() != d["epi"]).sum(1)
            t[op.output] = torch.clamp(lens, min=1)
        elif op.kind == K_ATTENTION:
            b, s, nh, hd = d["B"], d["S"], d["NH"], d["HD"]
            hid = nh * hd
            qkv = x.reshape(b, s, 3, nh, hd)
            q = qkv[:, :, 0].permute(0, 2, 1, 3)  # [B, NH, S, HD]
            k = qkv[:, :, 1].permute(0, 2, 1, 3)
            v = qkv[:, :, 2].permute(0, 2, 1, 3)
            scores = q @ k.transpose(-1, -2) * d["att_scale"]
            if len(op.inputs) > 1:  # varlen: mask right-padded keys
                lens = t[op.inputs[1]].long()
                keymask = torch.arange(s)[None, :] >= lens[:, None]  # [B,S]
                scores = scores.masked_fill(
                    keymask[:, None, None, :], float("-inf"))
            if d.get("causal"):
                cm = torch.arange(s)[None, :] > torch.arange(s)[:, None]
                scores = scores.masked_fill(cm[None, None], float("-inf"))
            att = torch.softmax(scores, dim=-1)
            y = (att @ v).permute(0, 2, 1, 3).reshape(b * s, hid)
            if d.get("epi") == 3:  # fused fp8 output
                y = _fp8_round(y / d["q_scale"])
            t[op.output] = y
        elif op.kind == K_RMSNORM:
            g_ = torch.from_numpy(op.scale)
            r = torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + d["eps"])
            t[op.output] = x * r * g_
        elif op.kind == K_SILU_MUL:
            t[op.output] = F.silu(x) * t[op.inputs[1]]
        elif op.kind == K_ROPE:
            m, n3 = x.shape
            heads, hd, seq = d["NH"], d["HD"], d["S"]
            half = hd // 2
            theta = d["eps"]
            posv = torch.arange(m) % seq
            dvec = torch.arange(half, dtype=torch.float64)
            ang = posv[:, None].double() * theta ** (-2.0 * dvec / hd)
            cos = torch.cos(ang).float()  # [m, half]
            sin = torch.sin(ang).float()
            y = x.clone()
            hid = heads * hd
            for blk in range(2):  # q then k; v untouched
                base = blk * hid
                v = x[:, base:base + hid].reshape(m, heads, hd)
                x0 = v[..., :half]
                x1 = v[..., half:]
                r0 = x0 * cos[:, None, :] - x1 * sin[:, None, :]
                r1 = x0 * sin[:, None, :] + x1 * cos[:, None, :]
                y[:, base:base + hid] = torch.cat([r0, r1], -1).reshape(
                    m, hid)
            t[op.output] = y
        elif op.kind == K_CLIP:
            t[op.output] = torch.clamp(x, d["res_scale"], d["q_scale"])
        elif op.kind == K_TRANSPOSE2D:
            t[op.output] = x.reshape(d["M"], d["N"]).t().contiguous()
        elif op.kind == K_COPY2D:
            rows, c, ldd, coff = d["M"], d["C"], d["Cout"], d["epi"]
            if op.output not in t:
                t[op.output] = torch.zeros(*plan.shapes[op.output])
            t[op.output].reshape(rows, ldd)[:, coff:coff + c] = \
                x.reshape(rows, c)
        else:
            raise ValueError(f"bad op kind {op.kind}")
    if return_all:
        return {k: v.numpy() for k, v in t.items()}
    return t[plan.output_name].numpy()
