"""TSAN/ASAN tier for the native IO engine (SURVEY §5 race-detection
mapping: the reference relies on Go's race detector + single-reconciler
discipline; here the threaded C++ layer gets real sanitizer builds).

Compiles kfio.cpp together with the C++ harness under -fsanitize=thread
and -fsanitize=address and runs the concurrent read/write paths; any
race or heap error aborts the binary with a nonzero exit."""
import os
import shutil
import subprocess

import pytest

CSRC = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "kubeflow_amd", "ops", "csrc_cpp")


def _build_and_run(tmp_path, sanitizer):
    exe = tmp_path / f"kfio_{sanitizer}"
    cmd = ["g++", "-O1", "-g", "-std=c++17", f"-fsanitize={sanitizer}",
           "-fno-omit-frame-pointer", "-pthread",
           os.path.join(CSRC, "kfio.cpp"),
           os.path.join(CSRC, "kfio_sanitize_test.cpp"),
           "-o", str(exe)]
    build = subprocess.run(cmd, capture_output=True, text=True, timeout=300)
    assert build.returncode == 0, build.stderr
    env = dict(os.environ)
    env["TSAN_OPTIONS"] = "halt_on_error=1"
    env["ASAN_OPTIONS"] = "detect_leaks=0"  # harness exits by return
    run = subprocess.run([str(exe), str(tmp_path)], capture_output=True,
                         text=True, timeout=300, env=env)
    assert run.returncode == 0, f"rc={run.returncode}\n{run.stdout}\n{run.stderr}"
    assert "OK" in run.stdout


@pytest.mark.skipif(shutil.which("g++") is None, reason="no g++")
def test_kfio_under_tsan(tmp_path):
    _build_and_run(tmp_path, "thread")


@pytest.mark.skipif(shutil.which("g++") is None, reason="no g++")
def test_kfio_under_asan(tmp_path):
    _build_and_run(tmp_path, "address")
