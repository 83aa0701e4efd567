"""Sanitizer jobs (the reference had none; adopt ASAN/TSAN):
build the host-side concurrency primitives (hybrid futex mutex/condvar)
with ThreadSanitizer and AddressSanitizer+UBSan and run them. Pure host
code — runs in CPU CI with plain g++."""
import subprocess
import sys
from pathlib import Path

import pytest

ROOT = Path(__file__).resolve().parent.parent
SRC = ROOT / "csrc" / "tests" / "sanitize_host.cpp"


def _build_and_run(tmp_path, flags, name):
    exe = tmp_path / name
    r = subprocess.run(
        ["g++", "-std=c++17", "-O1", "-g", *flags, str(SRC), "-o", str(exe),
         "-lpthread"], capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, f"build failed:\n{r.stderr}"
    r = subprocess.run([str(exe)], capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, f"{name} failed:\n{r.stdout}\n{r.stderr}"
    assert "SANITIZE_HOST_OK" in r.stdout
    # sanitizers report to stderr; any report is a failure
    assert "WARNING" not in r.stderr and "ERROR" not in r.stderr, r.stderr


def test_tsan_hybrid_primitives(tmp_path):
    _build_and_run(tmp_path, ["-fsanitize=thread"], "tsan_host")


def test_asan_ubsan_hybrid_primitives(tmp_path):
    _build_and_run(tmp_path, ["-fsanitize=address,undefined"], "asan_host")
