"""ONNX import/export round-trip tests (pure wire-format codec, CPU)."""
import numpy as np

from trtlab_amd.engine.onnx_io import export_onnx, import_onnx
from trtlab_amd.engine.planner import Planner
from trtlab_amd.engine.reference import run_reference
from trtlab_amd.models import build_resnet


def test_onnx_roundtrip_small_convnet():
    g = build_resnet(50, batch=1, image=64, seed=0, calibrate=False)
    data = export_onnx(g)
    assert len(data) > 1_000_000  # weights present
    g2 = import_onnx(data, name="roundtrip")

    plan1 = Planner().compile(g)
    plan2 = Planner().compile(g2)
    assert len(plan1.ops) == len(plan2.ops)

    x = np.random.RandomState(1).randn(1, 64, 64, 3).astype(np.float32) * 0.5
    out1 = run_reference(plan1, x)
    out2 = run_reference(plan2, x)
    assert np.allclose(out1, out2, atol=1e-4), np.abs(out1 - out2).max()


def test_onnx_import_batch_override():
    g = build_resnet(50, batch=1, image=64, seed=0, calibrate=False)
    data = export_onnx(g)
    g2 = import_onnx(data, batch=4)
    assert g2.tensors[g2.input_name].shape[0] == 4
    plan = Planner().compile(g2)
    assert plan.input_shape[0] == 4
