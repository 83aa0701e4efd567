"""DenseNet family: dense blocks with channel concatenation
(K_COPY2D lowering) + standalone pre-activation batchnorm (K_CHAFF),
validated against an independent plain-torch implementation."""
import numpy as np
import pytest
import torch
import torch.nn.functional as F

from trtlab_amd.engine.planner import K_CHAFF, K_COPY2D, Planner
from trtlab_amd.engine.reference import run_reference
from trtlab_amd.models.densenet import build_densenet

CFG = dict(batch=1, image=64, growth=16, blocks=(4, 4), init_ch=64,
           classes=10, seed=0)


def _torch_densenet(g, x):
    nodes = {n.name: n for n in g.nodes}

    def conv(v, name, stride=1, pad=0):
        wt = torch.from_numpy(nodes[name].attrs["weight"])
        return F.conv2d(v.permute(0, 3, 1, 2), wt, stride=stride,
                        padding=pad).permute(0, 2, 3, 1)

    def bn_relu(v, name):
        a = nodes[name + "_bn"].attrs
        s = a["gamma"] / np.sqrt(a["var"] + a["eps"])
        b = a["beta"] - a["mean"] * s
        return torch.relu(v * torch.from_numpy(s) + torch.from_numpy(b))

    v = torch.from_numpy(x)
    v = conv(v, "stem", stride=2, pad=3)
    v = bn_relu(v, "stem")
    v = F.max_pool2d(v.permute(0, 3, 1, 2), 3, 2, 1).permute(0, 2, 3, 1)
    blocks = CFG["blocks"]
    for bi, nl in enumerate(blocks):
        for li in range(nl):
            nm = f"b{bi}l{li}"
            y = bn_relu(v, nm + "_pre")
            y = conv(y, nm + "_c1")
            y = bn_relu(y, nm + "_mid")
            y = conv(y, nm + "_c3", pad=1)
            v = torch.cat([v, y], dim=-1)
        if bi + 1 < len(blocks):
            v = bn_relu(v, f"t{bi}")
            v = conv(v, f"t{bi}_conv")
            v = F.avg_pool2d(v.permute(0, 3, 1, 2), 2,
                             2).permute(0, 2, 3, 1)
    v = bn_relu(v, "final")
    v = v.mean(dim=(1, 2))
    a = nodes["head"].attrs
    return (v @ torch.from_numpy(a["weight"]).t() +
            torch.from_numpy(a["bias"])).numpy()


def test_densenet_reference_matches_torch_oracle():
    g = build_densenet(**CFG)
    plan = Planner().compile(g)
    kinds = [d["kind"] for d in plan.ops]
    # standalone (pre-act) BNs: one per dense layer (producer = concat/
    # pool) + one transition + the final — the mid/stem BNs follow convs
    # and fuse into their epilogues instead
    assert kinds.count(K_CHAFF) == 8 + 1 + 1
    assert kinds.count(K_COPY2D) == 16       # 2 per concat, 8 concats
    x = (np.random.RandomState(0).randn(*plan.input_shape) * 0.5).astype(
        np.float32)
    out = run_reference(plan, x)
    ref = _torch_densenet(g, x)
    err = np.abs(out - ref).max() / max(np.abs(ref).max(), 1e-6)
    assert err < 2e-3, err


@pytest.mark.gpu
def test_densenet_engine_matches_reference():
    """Dense blocks end-to-end on the captured engine: concat copies +
    channel-affine pre-activations + conv/pool/head."""
    from trtlab_amd.engine.runtime import NativeEngine

    g = build_densenet(**CFG)
    plan = Planner().compile(g)
    ctx = NativeEngine(plan).create_context(capture=True)
    x = (np.random.RandomState(3).randn(*plan.input_shape) * 0.5).astype(
        np.float32)
    out = ctx.infer(x).astype(np.float32)
    ref = run_reference(plan, x)
    err = np.abs(out - ref).max() / max(np.abs(ref).max(), 1e-6)
    assert err < 0.08, err


def test_densenet_onnx_roundtrip():
    """DenseNet exports (Conv/BN/Relu/Concat/AveragePool) and re-imports
    to an identical-output graph — including standalone (non-conv-fed)
    BatchNormalization, which lowers to the channel-affine op."""
    from trtlab_amd.engine.onnx_io import export_onnx, import_onnx

    g = build_densenet(**CFG)
    g2 = import_onnx(export_onnx(g))
    p1, p2 = Planner().compile(g), Planner().compile(g2)
    x = (np.random.RandomState(1).randn(*p1.input_shape) * 0.5).astype(
        np.float32)
    a, b = run_reference(p1, x), run_reference(p2, x)
    assert np.abs(a - b).max() / max(np.abs(a).max(), 1e-6) < 1e-5
