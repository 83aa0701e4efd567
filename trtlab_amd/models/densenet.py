"""DenseNet builder — dense blocks of BN+ReLU+conv layers whose outputs
CONCATENATE along channels (the torchvision DenseNet-BC shape:
bottleneck 1x1 -> 3x3, growth-rate channels per layer, transition
1x1 + avgpool between blocks).

Exercises the channel-concat lowering (K_COPY2D strided copies) inside a
real network — the reference's model zoo is ResNet-family only, so this
is breadth beyond it. Post-activation ordering differs from ResNet:
DenseNet applies BN+ReLU BEFORE each conv, which the planner folds as a
standalone batchnorm+relu ahead of the conv (BN+ReLU epilogues attach to
the PRODUCING conv; here the producer is a concat, so the pre-norm stays
a separate op — correctness first, the concat fusion is the cost).
"""
from __future__ import annotations

import numpy as np

from trtlab_amd.engine.ir import Graph


def build_densenet(batch: int = 8, image: int = 224, growth: int = 32,
                   blocks=(6, 12, 24, 16), init_ch: int = 64,
                   classes: int = 1000, seed: int = 0) -> Graph:
    """Default config = DenseNet-121 (blocks 6/12/24/16, growth 32)."""
    rng = np.random.RandomState(seed)

    def w(cout, cin, k):
        return (rng.randn(cout, cin, k, k) *
                np.sqrt(2.0 / (cin * k * k))).astype(np.float32)

    def bnp(c):
        return (rng.uniform(0.8, 1.2, c).astype(np.float32),
                (rng.randn(c) * 0.05).astype(np.float32),
                (rng.randn(c) * 0.1).astype(np.float32),
                rng.uniform(0.5, 1.5, c).astype(np.float32))

    def bn_relu(g, x, c, name):
        gm, bt, mu, var = bnp(c)
        x = g.batchnorm(x, gm, bt, mu, var, name=f"{name}_bn")
        return g.relu(x, name=f"{name}_relu")

    g = Graph(f"densenet_g{growth}_b{batch}")
    x = g.input((batch, image, image, 3), name="input")
    x = g.conv2d(x, w(init_ch, 3, 7), stride=2, padding=3, name="stem")
    x = bn_relu(g, x, init_ch, "stem")
    x = g.maxpool(x, kernel=3, stride=2, padding=1, name="stem_pool")

    ch = init_ch
    for bi, nlayers in enumerate(blocks):
        for li in range(nlayers):
            nm = f"b{bi}l{li}"
            # bottleneck: BN+ReLU -> 1x1 (4*growth) -> BN+ReLU -> 3x3
            y = bn_relu(g, x, ch, nm + "_pre")
            y = g.conv2d(y, w(4 * growth, ch, 1), name=nm + "_c1")
            y = bn_relu(g, y, 4 * growth, nm + "_mid")
            y = g.conv2d(y, w(growth, 4 * growth, 3), padding=1,
                         name=nm + "_c3")
            x = g.concat([x, y], name=nm + "_cat")
            ch += growth
        if bi + 1 < len(blocks):
            # transition: BN+ReLU -> 1x1 halve channels -> 2x2 avgpool
            x = bn_relu(g, x, ch, f"t{bi}")
            ch //= 2
            x = g.conv2d(x, w(ch, ch * 2, 1), name=f"t{bi}_conv")
            x = g.avgpool(x, kernel=2, stride=2, name=f"t{bi}_pool")
    x = bn_relu(g, x, ch, "final")
    x = g.global_avgpool(x, name="gap")
    g.gemm(x, (rng.randn(classes, ch) * np.sqrt(1.0 / ch))
           .astype(np.float32),
           (rng.randn(classes) * 0.02).astype(np.float32), name="head")
    return g
