"""Sanitizer pass over the native gather core.

Builds tests/native/gather_tsan.cpp with -fsanitize=thread and runs it:
TSAN instruments the concurrent factorize phases (per-thread tables,
shared code arrays, parallel remap); any data race fails the run.
"""
import shutil
import subprocess
import sys

import pytest

ROOT = __file__.rsplit("/", 2)[0]


@pytest.mark.skipif(shutil.which("g++") is None, reason="no g++")
def test_gather_core_under_tsan(tmp_path):
    exe = str(tmp_path / "gather_tsan")
    build = subprocess.run(
        ["g++", "-std=c++17", "-O1", "-g", "-fsanitize=thread", "-pthread",
         "-I", f"{ROOT}/mi355x_scale/groupby/csrc",
         f"{ROOT}/tests/native/gather_tsan.cpp", "-o", exe],
        capture_output=True, text=True, timeout=300)
    assert build.returncode == 0, build.stderr[-2000:]
    run = subprocess.run([exe], capture_output=True, text=True, timeout=300)
    sys.stderr.write(run.stdout + run.stderr)
    assert run.returncode == 0, run.stderr[-2000:]
    assert "OK" in run.stdout
    assert "WARNING: ThreadSanitizer" not in run.stderr
