"""BN statistics calibration for random-weight models.

A randomly-initialized ResNet with arbitrary BN running stats amplifies
variance through the residual stages (var roughly doubles per block) and
overflows fp16. Trained networks don't, because BN's running stats match
the actual activation statistics. We reproduce that property: one fp32
CPU forward on synthetic data, setting each batchnorm's mean/var to the
observed per-channel statistics of its input. The result is numerically
self-normalizing — matching the reference's random-weight synthetic-model
protocol (models/README.md:4-8) while staying fp16-safe.
"""
from __future__ import annotations

import numpy as np
import torch
import torch.nn.functional as F

from trtlab_amd.engine.ir import Graph


@torch.no_grad()
def calibrate_bn(g: Graph, sample: np.ndarray) -> None:
    """Node-level fp32 interpretation of the graph; rewrites BN attrs."""
    t = {g.input_name: torch.from_numpy(np.ascontiguousarray(sample)).float()}
    for n in g.nodes:
        if n.kind == "input":
            continue
        x = t[n.inputs[0]]
        if n.kind == "conv2d":
            w = torch.from_numpy(n.attrs["weight"])
            y = F.conv2d(x.permute(0, 3, 1, 2), w, stride=n.attrs["stride"],
                         padding=n.attrs["padding"]).permute(0, 2, 3, 1)
        elif n.kind == "batchnorm":
            flat = x.reshape(-1, x.shape[-1])
            mean = flat.mean(0)
            var = flat.var(0, unbiased=False).clamp_min(1e-3)
            n.attrs["mean"] = mean.numpy().astype(np.float32)
            n.attrs["var"] = var.numpy().astype(np.float32)
            a = n.attrs
            y = (x - mean) / torch.sqrt(var + a["eps"])
            y = y * torch.from_numpy(a["gamma"]) + torch.from_numpy(a["beta"])
        elif n.kind == "relu":
            y = F.relu(x)
        elif n.kind == "gelu":
            y = F.gelu(x, approximate="tanh")
        elif n.kind == "add":
            y = x + t[n.inputs[1]]
        elif n.kind == "maxpool":
            y = F.max_pool2d(x.permute(0, 3, 1, 2), n.attrs["kernel"],
                             stride=n.attrs["stride"],
                             padding=n.attrs["padding"]).permute(0, 2, 3, 1)
        elif n.kind == "gavgpool":
            nb, h, w_, c = x.shape
            y = x.reshape(nb, h * w_, c).mean(1)
        elif n.kind == "gemm":
            y = x @ torch.from_numpy(n.attrs["weight"]).t()
            if n.attrs.get("bias") is not None:
                y = y + torch.from_numpy(n.attrs["bias"])
        elif n.kind == "softmax":
            y = F.softmax(x, dim=-1)
        elif n.kind == "layernorm":
            y = F.layer_norm(x, (x.shape[-1],),
                             torch.from_numpy(n.attrs["gamma"]),
                             torch.from_numpy(n.attrs["beta"]), n.attrs["eps"])
        elif n.kind == "add_layernorm":
            s = x + t[n.inputs[1]]
            y = F.layer_norm(s, (s.shape[-1],),
                             torch.from_numpy(n.attrs["gamma"]),
                             torch.from_numpy(n.attrs["beta"]), n.attrs["eps"])
        elif n.kind == "attention":
            a = n.attrs
            b = x.shape[0] // a["seq"]
            qkv = x.reshape(b, a["seq"], 3, a["heads"], a["head_dim"])
            q, k, v = (qkv[:, :, i].permute(0, 2, 1, 3) for i in range(3))
            att = torch.softmax(
                q @ k.transpose(-1, -2) / np.sqrt(a["head_dim"]), dim=-1)
            y = (att @ v).permute(0, 2, 1, 3).reshape(x.shape[0], -1)
        else:
            raise ValueError(f"calibrate: unknown node kind {n.kind}")
        t[n.output] = y
