"""Self-contained torchvision-equivalent ResNet (Bottleneck) + an ONNX
exporter that reproduces torch.onnx.export's graph structure.

Purpose: the offline image has neither torchvision nor
the onnx package, so a literal stock `resnet50.onnx` cannot be downloaded
or torch-exported here. This module provides the same gate with an
INDEPENDENT oracle: a plain-torch ResNet-50 (architecture identical to
torchvision.models.resnet50) whose ONNX serialization — written with our
own protobuf wire writer — has the exact node inventory torch.onnx.export
emits for it (Conv with bias-less weights, standalone BatchNormalization
nodes, Relu, MaxPool, Add, GlobalAveragePool, Flatten, Gemm with
transB=1). The importer must consume that file unmodified and match the
torch module's forward.

Reference: examples/ONNX/resnet50/build.py:1-20 (the reference leaned on
the ONNX zoo the same way).
"""
from __future__ import annotations

import numpy as np
import torch
import torch.nn as nn

from trtlab_amd.engine import onnx_wire as w
from trtlab_amd.engine.onnx_io import (_attr_i, _attr_ints, _node,
                                       _tensor_bytes, _value_info,
                                       _GRAPH_INIT, _GRAPH_INPUT,
                                       _GRAPH_NAME, _GRAPH_NODE,
                                       _GRAPH_OUTPUT, _MODEL_GRAPH)


class Bottleneck(nn.Module):
    expansion = 4

    def __init__(self, inplanes, planes, stride=1, downsample=None):
        super().__init__()
        self.conv1 = nn.Conv2d(inplanes, planes, 1, bias=False)
        self.bn1 = nn.BatchNorm2d(planes)
        self.conv2 = nn.Conv2d(planes, planes, 3, stride=stride, padding=1,
                               bias=False)
        self.bn2 = nn.BatchNorm2d(planes)
        self.conv3 = nn.Conv2d(planes, planes * 4, 1, bias=False)
        self.bn3 = nn.BatchNorm2d(planes * 4)
        self.relu = nn.ReLU(inplace=False)
        self.downsample = downsample
        self.stride = stride

    def forward(self, x):
        identity = x
        out = self.relu(self.bn1(self.conv1(x)))
        out = self.relu(self.bn2(self.conv2(out)))
        out = self.bn3(self.conv3(out))
        if self.downsample is not None:
            identity = self.downsample(x)
        return self.relu(out + identity)


class TorchResNet(nn.Module):
    """torchvision.models.resnet50-equivalent (Bottleneck [3,4,6,3])."""

    def __init__(self, layers=(3, 4, 6, 3), num_classes=1000, seed=0):
        super().__init__()
        torch.manual_seed(seed)
        self.inplanes = 64
        self.conv1 = nn.Conv2d(3, 64, 7, stride=2, padding=3, bias=False)
        self.bn1 = nn.BatchNorm2d(64)
        self.relu = nn.ReLU(inplace=False)
        self.maxpool = nn.MaxPool2d(3, stride=2, padding=1)
        self.layer1 = self._make_layer(64, layers[0])
        self.layer2 = self._make_layer(128, layers[1], stride=2)
        self.layer3 = self._make_layer(256, layers[2], stride=2)
        self.layer4 = self._make_layer(512, layers[3], stride=2)
        self.avgpool = nn.AdaptiveAvgPool2d((1, 1))
        self.fc = nn.Linear(512 * 4, num_classes)
        # random running stats so BN actually transforms in eval mode
        g = torch.Generator().manual_seed(seed + 1)
        for m in self.modules():
            if isinstance(m, nn.BatchNorm2d):
                m.running_mean.copy_(torch.randn(m.num_features,
                                                 generator=g) * 0.1)
                m.running_var.copy_(torch.rand(m.num_features,
                                               generator=g) * 0.5 + 0.5)
                m.weight.data.copy_(torch.rand(m.num_features,
                                               generator=g) + 0.5)
                m.bias.data.copy_(torch.randn(m.num_features,
                                              generator=g) * 0.1)
        self.eval()

    def _make_layer(self, planes, blocks, stride=1):
        downsample = None
        if stride != 1 or self.inplanes != planes * 4:
            downsample = nn.Sequential(
                nn.Conv2d(self.inplanes, planes * 4, 1, stride=stride,
                          bias=False),
                nn.BatchNorm2d(planes * 4))
        layers = [Bottleneck(self.inplanes, planes, stride, downsample)]
        self.inplanes = planes * 4
        layers += [Bottleneck(self.inplanes, planes)
                   for _ in range(1, blocks)]
        return nn.Sequential(*layers)

    def forward(self, x):
        x = self.maxpool(self.relu(self.bn1(self.conv1(x))))
        x = self.layer4(self.layer3(self.layer2(self.layer1(x))))
        x = self.avgpool(x)
        x = torch.flatten(x, 1)
        return self.fc(x)


def export_resnet_onnx(model: TorchResNet, batch: int = 1,
                       image: int = 224) -> bytes:
    """Serialize `model` to ONNX bytes with torch.onnx.export's node
    inventory: Conv / BatchNormalization / Relu / MaxPool / Add /
    GlobalAveragePool / Flatten / Gemm(transB=1), NCHW I/O, raw-data
    initializers, slash-path node names."""
    inits: list = []
    nodes: list = []
    ctr = [0]

    def init(name: str, arr: torch.Tensor) -> str:
        inits.append(_tensor_bytes(name, arr.detach().numpy().astype(
            np.float32)))
        return name

    def fresh(prefix: str) -> str:
        ctr[0] += 1
        return f"{prefix}_{ctr[0]}"

    def conv(x, m: nn.Conv2d, path: str) -> str:
        wname = init(f"{path}.weight", m.weight)
        out = fresh("conv")
        nodes.append(_node("Conv", [x, wname], [out],
                           _attr_ints("dilations", [1, 1]),
                           _attr_i("group", 1),
                           _attr_ints("kernel_shape",
                                      list(m.kernel_size)),
                           _attr_ints("pads", list(m.padding) * 2),
                           _attr_ints("strides", list(m.stride))))
        return out

    def bn(x, m: nn.BatchNorm2d, path: str) -> str:
        names = [init(f"{path}.{k}", v) for k, v in (
            ("weight", m.weight), ("bias", m.bias),
            ("running_mean", m.running_mean),
            ("running_var", m.running_var))]
        out = fresh("bn")
        nodes.append(_node("BatchNormalization", [x] + names, [out]))
        return out

    def relu(x) -> str:
        out = fresh("relu")
        nodes.append(_node("Relu", [x], [out]))
        return out

    def bottleneck(x, m: Bottleneck, path: str) -> str:
        idn = x
        out = relu(bn(conv(x, m.conv1, f"{path}.conv1"), m.bn1,
                      f"{path}.bn1"))
        out = relu(bn(conv(out, m.conv2, f"{path}.conv2"), m.bn2,
                      f"{path}.bn2"))
        out = bn(conv(out, m.conv3, f"{path}.conv3"), m.bn3, f"{path}.bn3")
        if m.downsample is not None:
            idn = bn(conv(x, m.downsample[0], f"{path}.downsample.0"),
                     m.downsample[1], f"{path}.downsample.1")
        add = fresh("add")
        nodes.append(_node("Add", [out, idn], [add]))
        return relu(add)

    x = "input"
    out = relu(bn(conv(x, model.conv1, "conv1"), model.bn1, "bn1"))
    mp = fresh("maxpool")
    nodes.append(_node("MaxPool", [out], [mp],
                       _attr_ints("kernel_shape", [3, 3]),
                       _attr_ints("pads", [1, 1, 1, 1]),
                       _attr_ints("strides", [2, 2])))
    out = mp
    for li, layer in enumerate(
            (model.layer1, model.layer2, model.layer3, model.layer4), 1):
        for bi, block in enumerate(layer):
            out = bottleneck(out, block, f"layer{li}.{bi}")
    gap = fresh("gap")
    nodes.append(_node("GlobalAveragePool", [out], [gap]))
    flat = fresh("flatten")
    nodes.append(_node("Flatten", [gap], [flat], _attr_i("axis", 1)))
    wname = init("fc.weight", model.fc.weight)  # [1000, 2048]
    bname = init("fc.bias", model.fc.bias)
    nodes.append(_node("Gemm", [flat, wname, bname], ["output"],
                       _attr_i("transB", 1)))

    gparts = [w.f_string(_GRAPH_NAME, "torch_resnet")]
    gparts += nodes  # _node() already wraps with the GRAPH_NODE tag
    gparts += [w.f_bytes(_GRAPH_INIT, i) for i in inits]
    gparts.append(w.f_bytes(
        _GRAPH_INPUT, _value_info("input", [batch, 3, image, image])))
    gparts.append(w.f_bytes(
        _GRAPH_OUTPUT, _value_info("output", [batch, 1000])))
    graph = b"".join(gparts)
    return w.f_bytes(_MODEL_GRAPH, graph)
