"""CLI e2e: serve in a subprocess, drive with the CLI commands."""
import os
import signal
import subprocess
import sys
import time
import urllib.request

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.fixture(scope="module")
def served(tmp_path_factory):
    root = str(tmp_path_factory.mktemp("plat"))
    port = 18741
    env = dict(os.environ, PYTHONPATH=REPO, KF_API=f"http://127.0.0.1:{port}")
    proc = subprocess.Popen(
        [sys.executable, "-m", "kubeflow_amd.cli", "serve", "--root", root,
         "--port", str(port)], env=env,
        stdout=subprocess.PIPE, stderr=subprocess.STDOUT)
    deadline = time.time() + 60
    while time.time() < deadline:
        try:
            urllib.request.urlopen(f"http://127.0.0.1:{port}/healthz",
                                   timeout=1)
            break
        except Exception:
            if proc.poll() is not None:
                raise RuntimeError(proc.stdout.read().decode())
            time.sleep(0.3)
    yield env, port
    proc.send_signal(signal.SIGINT)
    try:
        proc.wait(timeout=10)
    except subprocess.TimeoutExpired:
        proc.kill()


def _cli(env, *args):
    return subprocess.run(
        [sys.executable, "-m", "kubeflow_amd.cli", *args], env=env,
        capture_output=True, text=True, timeout=120)


def test_cli_submit_get_logs(served):
    env, port = served
    r = _cli(env, "submit-train", "--model", "mnist-mlp", "--name", "cli-job",
             "--gpus", "1", "--steps", "3", "--micro-batch", "8")
    assert r.returncode == 0, r.stderr
    # on a CPU box gpus_per_replica=1 -> falls back to CPU via want_gpu logic
    deadline = time.time() + 120
    done = False
    while time.time() < deadline and not done:
        r = _cli(env, "get", "pytorchjobs")
        assert r.returncode == 0, r.stderr
        done = "Succeeded" in r.stdout
        time.sleep(1)
    assert done, r.stdout
    r = _cli(env, "status", "cli-job")
    assert '"Succeeded"' in r.stdout or "JobSucceeded" in r.stdout
    r = _cli(env, "logs", "cli-job")
    assert r.returncode == 0
    r = _cli(env, "events")
    assert "JobSucceeded" in r.stdout
    # dashboard: / serves the SPA shell; /classic renders the workloads
    with urllib.request.urlopen(f"http://127.0.0.1:{port}/", timeout=5) as h:
        page = h.read().decode()
    assert "kubeflow-amd" in page and "/static/app.js" in page
    with urllib.request.urlopen(f"http://127.0.0.1:{port}/classic",
                                timeout=5) as h:
        page = h.read().decode()
    assert "kubeflow-amd" in page and "PyTorchJob" in page


def test_cli_apply_manifest(served, tmp_path):
    env, port = served
    mf = tmp_path / "nb.yaml"
    mf.write_text("""
kind: Notebook
metadata:
  name: cli-nb
  namespace: default
spec:
  template:
    spec:
      containers:
      - image: kubeflow-amd/session:latest
""")
    r = _cli(env, "apply", "-f", str(mf))
    assert r.returncode == 0, r.stderr
    r = _cli(env, "get", "notebooks")
    assert "cli-nb" in r.stdout
    r = _cli(env, "delete", "notebooks", "cli-nb")
    assert r.returncode == 0
