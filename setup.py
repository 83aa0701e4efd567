"""trtlab_amd build: hipcc-driven in-tree extension for gfx950 (MI355X).

Builds trtlab_amd/_C.so directly with hipcc (no torch linkage — the runtime
is native; torch is only used at the Python API boundary). The .so is built
in-tree so it travels with repo snapshots.
"""
import os
import subprocess
import sys
import sysconfig
from concurrent.futures import ThreadPoolExecutor
from pathlib import Path

ROOT = Path(__file__).resolve().parent
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
ARCH = os.environ.get("TRTLAB_GPU_ARCH", "gfx950")

SOURCES = [
    "csrc/kernels/gemm.hip",
    "csrc/kernels/gemm_mx.hip",
    "csrc/kernels/decode.hip",
    "csrc/kernels/conv.hip",
    "csrc/kernels/splitk.hip",
    "csrc/kernels/pool.hip",
    "csrc/kernels/normalize.hip",
    "csrc/kernels/elementwise.hip",
    "csrc/kernels/attention.hip",
    "csrc/kernels/embedding.hip",
    "csrc/runtime/memory.cpp",
    "csrc/runtime/executor.cpp",
    "csrc/runtime/comm.cpp",
    "csrc/runtime/bfit.cpp",
    "csrc/ext.cpp",
]

HEADERS = [
    "csrc/common.h",
    "csrc/kernels/gemm_common.h",
    "csrc/kernels/launchers.h",
    "csrc/runtime/runtime.h",
    "csrc/runtime/comm.h",
]


def _pybind11_include():
    import pybind11
    return pybind11.get_include()


def _needs_build(src: Path, obj: Path) -> bool:
    if not obj.exists():
        return True
    newest_hdr = max((ROOT / h).stat().st_mtime for h in HEADERS)
    return obj.stat().st_mtime < max(src.stat().st_mtime, newest_hdr)


def build(verbose: bool = True) -> Path:
    out = ROOT / "trtlab_amd" / "_C.so"
    objdir = ROOT / "build" / "obj"
    objdir.mkdir(parents=True, exist_ok=True)

    py_inc = sysconfig.get_paths()["include"]
    common = [
        f"--offload-arch={ARCH}",
        "-O3",
        "-std=c++17",
        "-fPIC",
        f"-I{py_inc}",
        f"-I{_pybind11_include()}",
        f"-I{ROOT / 'csrc'}",
        "-D__HIP_PLATFORM_AMD__=1",
        "-fvisibility=hidden",
    ]

    objs = []
    jobs = []
    for s in SOURCES:
        src = ROOT / s
        obj = objdir / (s.replace("/", "_") + ".o")
        objs.append(obj)
        if _needs_build(src, obj):
            cmd = [HIPCC, *common, "-x", "hip", "-c", str(src), "-o", str(obj)]
            jobs.append((src, cmd))

    def run(job):
        src, cmd = job
        if verbose:
            print(f"[hipcc] {src.relative_to(ROOT)}", flush=True)
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"hipcc failed for {src}:\n{r.stdout}\n{r.stderr}")

    with ThreadPoolExecutor(max_workers=os.cpu_count() or 4) as ex:
        list(ex.map(run, jobs))

    if jobs or not out.exists():
        link = [HIPCC, "-shared", "-fPIC", *[str(o) for o in objs],
                "-L/opt/rocm/lib", "-lrccl", "-o", str(out)]
        if verbose:
            print(f"[link] {out.relative_to(ROOT)}", flush=True)
        r = subprocess.run(link, capture_output=True, text=True)
        if r.returncode != 0:
            raise RuntimeError(f"link failed:\n{r.stdout}\n{r.stderr}")
    return out


if __name__ == "__main__":
    if len(sys.argv) > 1 and sys.argv[1] == "build_ext":
        build()
    else:
        build()
