"""trtlab_amd.engine — the TensorRT replacement.

Model IR -> fusion planner -> activation-arena memory plan -> native
graph-captured executor over hand-written CDNA4 kernels.
(Reference layer: trtlab/tensorrt — Runtime/Model/ExecutionContext/
Workspace/InferenceManager)
"""
from trtlab_amd.engine.ir import Graph, Node  # noqa: F401
from trtlab_amd.engine.planner import Planner, EnginePlan  # noqa: F401
from trtlab_amd.engine.reference import run_reference  # noqa: F401
