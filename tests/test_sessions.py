"""Notebook / Tensorboard / PVC controller tests (session processes)."""
import json
import os
import time
import urllib.request

import pytest

from kubeflow_amd.api import new_object, NotFoundError
from kubeflow_amd.api.objects import has_condition
from kubeflow_amd.platform import Platform


def _wait(pred, timeout=60, period=0.25):
    deadline = time.time() + timeout
    while time.time() < deadline:
        v = pred()
        if v:
            return v
        time.sleep(period)
    raise AssertionError("condition not reached")


def test_notebook_lifecycle_and_culler_contract(tmp_path):
    with Platform(root_dir=str(tmp_path)) as plat:
        nb = new_object("Notebook", "my-nb", "alice", spec={
            "template": {"spec": {"containers": [
                {"image": "kubeflow-amd/session:latest"}]}}},
            api_version="kubeflow.org/v1beta1")
        plat.store.create(nb)
        obj = _wait(lambda: (lambda o: o if o["status"].get("readyReplicas")
                             else None)(
            plat.store.get("Notebook", "my-nb", "alice")))
        url = obj["status"]["url"]
        assert url.endswith("/notebook/alice/my-nb/")
        # the culler contract: /api/status with last_activity
        with urllib.request.urlopen(
                url.rstrip("/") + "/api/status", timeout=5) as r:
            data = json.loads(r.read())
        assert "last_activity" in data

        # code execution endpoint works (the "kernel")
        body = json.dumps({"code": "print(2 + 3)"}).encode()
        req = urllib.request.Request(
            url.rstrip("/") + "/api/execute", data=body,
            headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=10) as r:
            out = json.loads(r.read())
        assert out["output"].strip() == "5"

        # stop via annotation -> replicas 0 (stop/start parity)
        plat.store.patch("Notebook", "my-nb", "alice", {
            "metadata": {"annotations": {
                "kubeflow-resource-stopped": "now"}}})
        _wait(lambda: plat.store.get("Notebook", "my-nb", "alice")
              ["status"].get("readyReplicas") == 0)
        # start again
        plat.store.patch("Notebook", "my-nb", "alice", {
            "metadata": {"annotations": {"kubeflow-resource-stopped": None}}})
        _wait(lambda: plat.store.get("Notebook", "my-nb", "alice")
              ["status"].get("readyReplicas") == 1)
        plat.store.delete("Notebook", "my-nb", "alice")


def test_pvc_bound_and_tensorboard_logspath(tmp_path):
    with Platform(root_dir=str(tmp_path)) as plat:
        pvc = new_object("PersistentVolumeClaim", "data", "alice", spec={
            "accessModes": ["ReadWriteOnce"],
            "resources": {"requests": {"storage": "5Gi"}}}, api_version="v1")
        plat.store.create(pvc)
        obj = _wait(lambda: (lambda o: o if o["status"].get("phase") == "Bound"
                             else None)(
            plat.store.get("PersistentVolumeClaim", "data", "alice")))
        path = obj["status"]["hostPath"]
        assert os.path.isdir(path)
        # drop a metrics file where tensorboard will scan
        rundir = os.path.join(path, "logs", "run1")
        os.makedirs(rundir)
        with open(os.path.join(rundir, "status.json"), "w") as f:
            json.dump({"step": 3, "metrics": {"loss": 0.5}}, f)

        tb = new_object("Tensorboard", "tb1", "alice",
                        spec={"logspath": "pvc://data/logs"},
                        api_version="tensorboard.kubeflow.org/v1alpha1")
        plat.store.create(tb)
        obj = _wait(lambda: (lambda o: o if has_condition(o, "Running")
                             else None)(
            plat.store.get("Tensorboard", "tb1", "alice")))
        # the viewer serves the scalar series from the PVC logdir
        sess = plat.tensorboard.sessions[obj["metadata"]["uid"]]
        with urllib.request.urlopen(
                f"http://127.0.0.1:{sess[1]}/data/runs", timeout=5) as r:
            runs = json.loads(r.read())["runs"]
        assert "run1" in runs

        # PVC deletion reclaims the directory
        plat.store.delete("PersistentVolumeClaim", "data", "alice")
        _wait(lambda: not os.path.isdir(path), timeout=20)


def test_tensorboard_rejects_cloud_paths(tmp_path):
    with Platform(root_dir=str(tmp_path)) as plat:
        tb = new_object("Tensorboard", "bad", "ns",
                        spec={"logspath": "gs://bucket/x"})
        plat.store.create(tb)
        obj = _wait(lambda: (lambda o: o if has_condition(o, "Failed")
                             else None)(
            plat.store.get("Tensorboard", "bad", "ns")))
        assert "unsupported" in [c for c in obj["status"]["conditions"]
                                 if c["type"] == "Failed"][0]["message"]


def test_culler_stops_idle_notebook(tmp_path, monkeypatch):
    """ENABLE_CULLING with a tiny IDLE_TIME stops an idle session
    (culler.go parity: poll /api/status, set stop annotation)."""
    monkeypatch.setenv("ENABLE_CULLING", "true")
    monkeypatch.setenv("IDLE_TIME", "0.0001")          # ~6 ms idle budget
    monkeypatch.setenv("CULLING_CHECK_PERIOD", "0.01")  # 0.6 s requeue
    with Platform(root_dir=str(tmp_path)) as plat:
        assert plat.notebook.enable_culling
        plat.store.create(new_object("Notebook", "idle-nb", "ns", spec={}))
        # NOTE: with a ~6 ms idle budget the ready (readyReplicas==1) state
        # can be culled away between store polls — don't wait on it, it is
        # incidental; the invariant under test is the cull itself.
        def culled():
            o = plat.store.get("Notebook", "idle-nb", "ns")
            ann = o["metadata"].get("annotations", {})
            return o if ("kubeflow-resource-stopped" in ann
                         and o["status"].get("readyReplicas") == 0) else None
        obj = _wait(culled, timeout=120)
        reasons = {e["reason"] for e in plat.store.events_for(obj)}
        assert "Culling" in reasons


def test_notebook_poddefault_env(tmp_path):
    """PodDefault env reaches notebook session processes too (the webhook
    mutates every pod in the reference)."""
    with Platform(root_dir=str(tmp_path)) as plat:
        pd = new_object("PodDefault", "nb-env", "ns", spec={
            "selector": {"matchLabels": {"wants-env": "yes"}},
            "env": [{"name": "KF_NB_INJECTED", "value": "hello"}]},
            api_version="kubeflow.org/v1alpha1")
        plat.store.create(pd)
        nb = new_object("Notebook", "envy", "ns", spec={},
                        labels={"wants-env": "yes"})
        plat.store.create(nb)
        obj = _wait(lambda: (lambda o: o if o["status"].get("readyReplicas")
                             else None)(
            plat.store.get("Notebook", "envy", "ns")), timeout=60)
        # session process can see the injected env via /api/execute
        url = obj["status"]["url"].rstrip("/")
        body = json.dumps({"code": "import os; print(os.environ.get('KF_NB_INJECTED'))"}).encode()
        req = urllib.request.Request(url + "/api/execute", data=body,
                                     headers={"Content-Type": "application/json"})
        with urllib.request.urlopen(req, timeout=10) as r:
            out = json.loads(r.read())
        assert out["output"].strip() == "hello"


def test_pvc_snapshot_and_restore(tmp_path):
    """rok-flavor parity: snapshot a PVC, restore a new PVC from the
    rok:// origin annotation (apps/rok/routes/post.py seam)."""
    with Platform(root_dir=str(tmp_path)) as plat:
        plat.store.create(new_object("PersistentVolumeClaim", "src", "ns",
                                     spec={}, api_version="v1"))
        obj = _wait(lambda: (lambda o: o if o["status"].get("phase") == "Bound"
                             else None)(
            plat.store.get("PersistentVolumeClaim", "src", "ns")))
        with open(os.path.join(obj["status"]["hostPath"], "data.txt"),
                  "w") as f:
            f.write("precious")
        url = plat.volume.snapshot("ns", "src")
        assert url.startswith("rok://")
        restored = new_object("PersistentVolumeClaim", "copy", "ns", spec={},
                              api_version="v1",
                              annotations={"rok/origin": url})
        plat.store.create(restored)
        obj = _wait(lambda: (lambda o: o if o["status"].get("phase") == "Bound"
                             else None)(
            plat.store.get("PersistentVolumeClaim", "copy", "ns")))
        with open(os.path.join(obj["status"]["hostPath"], "data.txt")) as f:
            assert f.read() == "precious"
        evs = {e["reason"] for e in plat.store.events_for(obj)}
        assert "SnapshotRestored" in evs


# ------------------------------------------------- notebook CRD versions

def _v1_payload(name):
    return {
        "apiVersion": "kubeflow.org/v1", "kind": "Notebook",
        "metadata": {"name": name, "namespace": "default"},
        "spec": {"template": {"spec": {"containers": [{
            "name": name, "image": "img:v1",
            "resources": {"limits": {"amd.com/gpu": 2, "cpu": "4"},
                          "requests": {"cpu": "4", "memory": "8Gi"}},
            "env": [{"name": "A", "value": "1"}],
        }]}}},
        "status": {"conditions": [
            {"type": "Running", "status": "True", "reason": "x",
             "message": "m", "bogusField": "dropped"}]},
    }


def test_normalize_v1_pod_template():
    from kubeflow_amd.controllers.notebook import NotebookReconciler
    out = NotebookReconciler.normalize(_v1_payload("nb"))
    assert out["apiVersion"] == "kubeflow.org/v1beta1"
    assert out["metadata"]["annotations"][
        "notebooks.kubeflow.org/original-api-version"] == "kubeflow.org/v1"
    assert out["spec"]["image"] == "img:v1"
    assert out["spec"]["gpus"] == 2
    assert out["spec"]["memory"] == "8Gi"
    assert out["spec"]["env"] == {"A": "1"}
    # conditions convert field-by-field; unknown keys drop
    c = out["status"]["conditions"][0]
    assert c["type"] == "Running" and "bogusField" not in c


def test_normalize_v1alpha1_pod_template():
    from kubeflow_amd.controllers.notebook import NotebookReconciler
    p = _v1_payload("nb2")
    p["apiVersion"] = "kubeflow.org/v1alpha1"
    out = NotebookReconciler.normalize(p)
    assert out["apiVersion"] == "kubeflow.org/v1beta1"
    assert out["spec"]["image"] == "img:v1"


def test_normalize_flat_spawner_shape():
    from kubeflow_amd.controllers.notebook import NotebookReconciler
    out = NotebookReconciler.normalize({
        "apiVersion": "kubeflow.org/v1beta1", "kind": "Notebook",
        "metadata": {"name": "nb3", "namespace": "default"},
        "spec": {"image": "img:flat", "gpus": 1, "cpu": "2",
                 "memory": "4Gi", "env": {"B": "2"}},
        "status": {},
    })
    c0 = out["spec"]["template"]["spec"]["containers"][0]
    assert c0["image"] == "img:flat"
    assert c0["resources"]["limits"]["amd.com/gpu"] == 1
    assert c0["env"] == [{"name": "B", "value": "2"}]
    # hub version untouched -> no original-version annotation
    assert "notebooks.kubeflow.org/original-api-version" not in \
        out["metadata"].get("annotations", {})


def test_notebook_gpu_session_allocates_and_releases(tmp_path, monkeypatch):
    """GPU notebooks flow through GangScheduler.allocate(exclusive=False)
    with HBM accounting: the session env pins
    HIP_VISIBLE_DEVICES and stopping the notebook releases the GPUs."""
    import time as _t

    from kubeflow_amd.api import new_object
    from kubeflow_amd.platform import Platform

    monkeypatch.setenv("KF_FAKE_GPUS", "2")
    with Platform(root_dir=str(tmp_path)) as plat:
        nb = new_object("Notebook", "gpu-nb", "default",
                        spec={"image": "kubeflow-amd/session:latest",
                              "gpus": 1, "gpuMemory": "16Gi"},
                        api_version="kubeflow.org/v1beta1")
        plat.store.create(nb)
        deadline = _t.time() + 60
        while _t.time() < deadline:
            obj = plat.store.get("Notebook", "gpu-nb", "default")
            if obj["status"].get("readyReplicas"):
                break
            _t.sleep(0.3)
        assert obj["status"].get("readyReplicas") == 1, obj["status"]
        assert plat.scheduler.ns_gpu_usage("default") == 1
        # stop via the annotation -> GPUs released
        obj["metadata"]["annotations"]["kubeflow-resource-stopped"] = "now"
        plat.store.update(obj, check_version=False)
        deadline = _t.time() + 30
        while _t.time() < deadline:
            if plat.scheduler.ns_gpu_usage("default") == 0:
                break
            _t.sleep(0.3)
        assert plat.scheduler.ns_gpu_usage("default") == 0
