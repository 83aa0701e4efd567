"""trtlab_amd — MI355X-native asynchronous GPU inference serving framework.

A from-scratch redesign of NVIDIA/tensorrt-laboratory's capabilities for
AMD Instinct MI355X (gfx950/CDNA4): native HIP kernels + graph executor,
pooled memory, async gRPC services, RCCL multi-GPU replicas.
"""

__version__ = "0.1.0"

# Import torch (if present) BEFORE the native module so a single HIP runtime
# (torch's bundled libamdhip64.so.7) is loaded for the whole process; our .so
# links the same SONAME and binds to it.
try:
    import torch  # noqa: F401
    _HAS_TORCH = True
except Exception:  # pragma: no cover
    _HAS_TORCH = False

try:
    from trtlab_amd import _C  # noqa: F401
    _HAS_NATIVE = True
    _NATIVE_ERR = None
except Exception as e:  # pragma: no cover
    _C = None
    _HAS_NATIVE = False
    _NATIVE_ERR = e


def native():
    """Return the native module, failing loudly if it is missing.

    On a GPU box the HIP path must be the one that runs — a silent
    eager/PyTorch fallback is a bug, not a feature.
    """
    if not _HAS_NATIVE:
        raise ImportError(
            f"trtlab_amd native extension (_C.so) is not built: {_NATIVE_ERR}. "
            "Run `python setup.py build_ext` first."
        )
    return _C


def has_gpu() -> bool:
    if not _HAS_NATIVE:
        return False
    try:
        return _C.hip.device_count() > 0
    except Exception:
        return False


from trtlab_amd import utils  # noqa: E402,F401
from trtlab_amd import memory  # noqa: E402,F401
from trtlab_amd import core  # noqa: E402,F401
