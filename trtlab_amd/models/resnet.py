"""ResNet-50/152 IR builders (NHWC, fp16 compute), random weights.

The reference serves ResNet-50/152 TensorRT engines built from caffe/ONNX
models (reference models/ + examples/ONNX/resnet50); BASELINE configs 2-4
are ResNet on MI355X. Weights are random-init (synthetic protocol).
"""
from __future__ import annotations

import numpy as np

from trtlab_amd.engine.ir import Graph

_LAYERS = {50: (3, 4, 6, 3), 101: (3, 4, 23, 3), 152: (3, 8, 36, 3)}
# basic-block depths (two 3x3 convs per block, expansion 1)
_LAYERS_BASIC = {18: (2, 2, 2, 2), 34: (3, 4, 6, 3)}


class _Init:
    def __init__(self, seed: int):
        self.rng = np.random.RandomState(seed)

    def conv(self, cout, cin, kh, kw):
        fan_in = cin * kh * kw
        return (self.rng.randn(cout, cin, kh, kw) *
                np.sqrt(2.0 / fan_in)).astype(np.float32)

    def bn(self, c):
        return dict(
            gamma=self.rng.uniform(0.8, 1.2, c).astype(np.float32),
            beta=self.rng.uniform(-0.1, 0.1, c).astype(np.float32),
            mean=(self.rng.randn(c) * 0.1).astype(np.float32),
            var=self.rng.uniform(0.5, 1.5, c).astype(np.float32),
        )

    def fc(self, nout, nin):
        w = (self.rng.randn(nout, nin) * np.sqrt(1.0 / nin)).astype(np.float32)
        b = (self.rng.randn(nout) * 0.01).astype(np.float32)
        return w, b


def _bottleneck(g: Graph, init: _Init, x: str, cin: int, mid: int,
                stride: int) -> str:
    cout = mid * 4
    # downsample branch FIRST so the planner fuses add+relu into conv3
    if stride != 1 or cin != cout:
        ds = g.conv2d(x, init.conv(cout, cin, 1, 1), stride=stride)
        ds = g.batchnorm(ds, **init.bn(cout))
        residual = ds
    else:
        residual = x
    h = g.conv2d(x, init.conv(mid, cin, 1, 1))
    h = g.batchnorm(h, **init.bn(mid))
    h = g.relu(h)
    h = g.conv2d(h, init.conv(mid, mid, 3, 3), stride=stride, padding=1)
    h = g.batchnorm(h, **init.bn(mid))
    h = g.relu(h)
    h = g.conv2d(h, init.conv(cout, mid, 1, 1))
    h = g.batchnorm(h, **init.bn(cout))
    h = g.add(h, residual)
    return g.relu(h)


def _basic_block(g: Graph, init: _Init, x: str, cin: int, mid: int,
                 stride: int) -> str:
    """ResNet-18/34 basic block: 3x3 -> 3x3 with identity/1x1 shortcut
    (torchvision BasicBlock). Downsample emitted first so the planner
    fuses the residual add + relu into the second conv's epilogue."""
    if stride != 1 or cin != mid:
        ds = g.conv2d(x, init.conv(mid, cin, 1, 1), stride=stride)
        ds = g.batchnorm(ds, **init.bn(mid))
        residual = ds
    else:
        residual = x
    h = g.conv2d(x, init.conv(mid, cin, 3, 3), stride=stride, padding=1)
    h = g.batchnorm(h, **init.bn(mid))
    h = g.relu(h)
    h = g.conv2d(h, init.conv(mid, mid, 3, 3), padding=1)
    h = g.batchnorm(h, **init.bn(mid))
    h = g.add(h, residual)
    return g.relu(h)


def build_resnet(depth: int = 50, batch: int = 8, image: int = 224,
                 classes: int = 1000, seed: int = 0, softmax: bool = False,
                 calibrate: bool = True) -> Graph:
    if depth not in _LAYERS and depth not in _LAYERS_BASIC:
        raise ValueError(f"unsupported resnet depth {depth}")
    basic = depth in _LAYERS_BASIC
    blocks = _LAYERS_BASIC[depth] if basic else _LAYERS[depth]
    init = _Init(seed)
    g = Graph(f"resnet{depth}_b{batch}")
    x = g.input((batch, image, image, 3))
    h = g.conv2d(x, init.conv(64, 3, 7, 7), stride=2, padding=3)
    h = g.batchnorm(h, **init.bn(64))
    h = g.relu(h)
    h = g.maxpool(h, kernel=3, stride=2, padding=1)
    cin = 64
    for stage, nblocks in enumerate(blocks):
        mid = 64 * (2 ** stage)
        for b in range(nblocks):
            stride = 2 if (stage > 0 and b == 0) else 1
            if basic:
                h = _basic_block(g, init, h, cin, mid, stride)
                cin = mid
            else:
                h = _bottleneck(g, init, h, cin, mid, stride)
                cin = mid * 4
    h = g.global_avgpool(h)
    w, bias = init.fc(classes, cin)
    h = g.gemm(h, w, bias)
    if softmax:
        h = g.softmax(h)
    if calibrate:
        from trtlab_amd.engine.calibrate import calibrate_bn

        sample = (init.rng.randn(min(batch, 2), image, image, 3)
                  .astype(np.float32) * 0.5)
        calibrate_bn(g, sample)
    return g
