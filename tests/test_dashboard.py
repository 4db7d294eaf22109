"""Dashboard SPA: asset serving, the UI's network contract, and the
ExponentialBackoff poller (run under node, the same engine class browsers
use — no browser binary exists in the CI container, so the browser-level
flows are covered by driving the SAME endpoint sequences the page code
issues, CSRF double-submit included)."""
import json
import os
import shutil
import subprocess

import pytest
from fastapi.testclient import TestClient

from kubeflow_amd.api.server import build_app
from kubeflow_amd.api.store import ObjectStore

STATIC = os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "kubeflow_amd", "dashboard", "static")


def client():
    return TestClient(build_app(ObjectStore()))


def test_spa_assets_served():
    c = client()
    r = c.get("/")
    assert r.status_code == 200
    assert 'src="/static/app.js"' in r.text
    for fname, ctype in (("app.js", "text/javascript"),
                         ("backend.js", "text/javascript"),
                         ("polling.js", "text/javascript"),
                         ("components.js", "text/javascript"),
                         ("pages.js", "text/javascript"),
                         ("style.css", "text/css")):
        r = c.get(f"/static/{fname}")
        assert r.status_code == 200, fname
        assert ctype in r.headers["content-type"], fname
    # path traversal stays inside the static dir
    r = c.get("/static/..%2F..%2Fconfig.py")
    assert r.status_code in (404, 200) and "DEFAULTS" not in r.text


def test_spa_routes_have_pages():
    """Every nav route in app.js has a page factory and every fetch path
    used by pages.js exists on the server (route-contract check)."""
    app_js = open(os.path.join(STATIC, "app.js")).read()
    pages_js = open(os.path.join(STATIC, "pages.js")).read()
    for name in ("homePage", "notebooksPage", "notebookSpawnerPage",
                 "volumesPage", "tensorboardsPage", "jobsPage",
                 "jobLogsPage", "servingPage", "experimentsPage",
                 "pipelinesPage", "activitiesPage", "workgroupPage"):
        assert name in app_js and f"export function {name}" in pages_js
    c = client()
    for path in ("/api/namespaces", "/api/config", "/api/dashboard-links",
                 "/api/metrics/node", "/api/activities/default",
                 "/api/workgroup/exists", "/api/workgroup/env-info"):
        r = c.get(path)
        assert r.status_code == 200, path
        assert r.json()["success"] is True, path


def test_ui_notebook_flow_with_csrf(monkeypatch, tmp_path):
    """The spawner/stop/delete sequence exactly as backend.js issues it:
    short-form POST, PATCH {stopped}, DELETE — with the double-submit
    CSRF cookie/header pair the SPA carries."""
    monkeypatch.setenv("KF_CSRF", "1")
    c = TestClient(build_app(ObjectStore()))
    # first GET sets the cookie (envelope middleware)
    r = c.get("/api/namespaces")
    tok = r.cookies.get("XSRF-TOKEN")
    assert tok
    hdr = {"X-XSRF-TOKEN": tok}
    # without the header the mutation is rejected
    r = c.post("/api/namespaces/default/notebooks", json={"name": "nb-ui"})
    assert r.status_code == 403 and r.json()["success"] is False
    # spawner submit (short form, like notebookSpawnerPage)
    r = c.post("/api/namespaces/default/notebooks",
               json={"name": "nb-ui", "image": "kubeflow-amd/session:latest",
                     "cpu": "2", "memory": "4Gi", "gpus": 0}, headers=hdr)
    assert r.json()["success"] is True
    # list shows it (notebooksPage poll)
    r = c.get("/api/namespaces/default/notebooks")
    names = [o["metadata"]["name"] for o in r.json()["notebooks"]]
    assert "nb-ui" in names
    # stop via PATCH {stopped: true} (stopAction)
    r = c.patch("/api/namespaces/default/notebooks/nb-ui",
                json={"stopped": True}, headers=hdr)
    assert r.json()["success"] is True
    r = c.get("/api/namespaces/default/notebooks/nb-ui")
    nb = r.json()["notebook"]
    assert "kubeflow-resource-stopped" in nb["metadata"]["annotations"]
    # delete (deleteAction)
    r = c.request("DELETE", "/api/namespaces/default/notebooks/nb-ui",
                  headers=hdr)
    assert r.json()["success"] is True


def test_ui_pytorchjob_submit_shape():
    """jobsPage's submit posts a full PyTorchJob object; the server must
    accept it and the list must round-trip the replica spec."""
    c = client()
    r = c.post("/api/namespaces/default/pytorchjobs", json={
        "apiVersion": "kubeflow.org/v1", "kind": "PyTorchJob",
        "metadata": {"name": "ui-job", "namespace": "default"},
        "spec": {"pytorchReplicaSpecs": {"Worker": {
            "replicas": 2, "restartPolicy": "Never",
            "template": {"model": "mnist-mlp", "steps": 5,
                         "gpus_per_replica": 0}}}},
    })
    assert r.json()["success"] is True
    r = c.get("/api/namespaces/default/pytorchjobs")
    jobs = r.json()["pytorchjobs"]
    assert jobs[0]["spec"]["pytorchReplicaSpecs"]["Worker"]["replicas"] == 2


@pytest.mark.skipif(shutil.which("node") is None, reason="no node")
def test_exponential_backoff_under_node(tmp_path):
    """Run polling.js in node: fast polling for `retries` ticks, doubling
    to maxInterval, reset-on-change returns to fast — the reference's
    exponential-backoff.ts:16-70 semantics."""
    harness = tmp_path / "t.mjs"
    shutil.copy(os.path.join(STATIC, "polling.js"), tmp_path / "polling.mjs")
    harness.write_text("""
import { ExponentialBackoff, changed } from "./polling.mjs";
const b = new ExponentialBackoff(async () => {}, {interval: 100, retries: 3, maxInterval: 1600});
const seq = [];
for (let i = 0; i < 20; i++) { seq.push(b.currentInterval()); b._n++; }
b.reset();
const afterReset = b.currentInterval();
console.log(JSON.stringify({seq, afterReset,
  chg: [changed({a:1},{a:2}), changed({a:1},{a:1})]}));
""")
    out = subprocess.run(["node", str(harness)], capture_output=True,
                         text=True, cwd=tmp_path, timeout=60)
    assert out.returncode == 0, out.stderr
    data = json.loads(out.stdout)
    assert data["seq"][:3] == [100, 100, 100]        # fast window
    assert data["seq"][3:6] == [200, 200, 200]       # first doubling
    assert max(data["seq"]) == 1600                  # capped
    assert data["seq"][-1] == 1600
    assert data["afterReset"] == 100                 # reset-on-change
    assert data["chg"] == [True, False]


@pytest.mark.skipif(shutil.which("node") is None, reason="no node")
def test_spa_modules_parse_under_node(tmp_path):
    for f in ("app.js", "backend.js", "polling.js", "components.js",
              "pages.js"):
        tgt = tmp_path / (f.replace(".js", ".mjs"))
        shutil.copy(os.path.join(STATIC, f), tgt)
        out = subprocess.run(["node", "--check", str(tgt)],
                             capture_output=True, text=True, timeout=60)
        assert out.returncode == 0, f"{f}: {out.stderr}"
