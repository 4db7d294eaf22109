"""Control-plane restart safety with a persistent store.

The reference's controllers are restart-safe because reconcile is
level-triggered over the apiserver's durable state (informer
initial list). Same property here: Platform(persist=True) reloads the JSONL
store, the manager enqueues every existing object, and:
  * terminal jobs are left alone (no relaunch),
  * a job that was mid-flight when the control plane died is relaunched.
"""
import time

from kubeflow_amd.api import new_object
from kubeflow_amd.api.objects import has_condition
from kubeflow_amd.platform import Platform


def _mk_job(name, steps):
    return new_object("PyTorchJob", name, "default", spec={
        "pytorchReplicaSpecs": {"Worker": {
            "replicas": 1, "restartPolicy": "Never",
            "template": {"model": "mnist-mlp", "steps": steps,
                         "micro_batch": 8, "gpus_per_replica": 0,
                         "status_every": 5, "save_final": False}}}})


def _wait_cond(store, kind, name, cond, timeout=120):
    deadline = time.time() + timeout
    while time.time() < deadline:
        obj = store.get(kind, name, "default")
        if has_condition(obj, cond):
            return obj
        assert not has_condition(obj, "Failed"), obj["status"]
        time.sleep(0.3)
    raise AssertionError(f"{name}: no {cond}: {obj['status']}")


def test_restart_leaves_terminal_jobs_alone(tmp_path):
    root = str(tmp_path)
    with Platform(root_dir=root, persist=True) as plat:
        plat.store.create(_mk_job("done-job", steps=4))
        _wait_cond(plat.store, "PyTorchJob", "done-job", "Succeeded")
    # control plane restarts against the same durable store
    with Platform(root_dir=root, persist=True) as plat2:
        obj = plat2.store.get("PyTorchJob", "done-job", "default")
        assert has_condition(obj, "Succeeded")  # state survived
        time.sleep(2.5)  # give the reconciler time to (wrongly) relaunch
        assert not plat2.pytorchjob.gangs  # it did not
        creates = [e for e in plat2.store.events_for(obj)
                   if e["reason"] == "SuccessfulCreate"]
        assert len(creates) == 1  # only the original launch


def test_restart_relaunches_midflight_job(tmp_path):
    root = str(tmp_path)
    with Platform(root_dir=root, persist=True) as plat:
        plat.store.create(_mk_job("long-job", steps=2000))
        _wait_cond(plat.store, "PyTorchJob", "long-job", "Running")
        time.sleep(1.0)  # let it get genuinely mid-flight
        obj = plat.store.get("PyTorchJob", "long-job", "default")
        assert not has_condition(obj, "Succeeded")
    # stop() killed the gang; the store still says Running.
    with Platform(root_dir=root, persist=True) as plat2:
        deadline = time.time() + 60
        while time.time() < deadline:
            obj = plat2.store.get("PyTorchJob", "long-job", "default")
            creates = [e for e in plat2.store.events_for(obj)
                       if e["reason"] == "SuccessfulCreate"]
            if len(creates) >= 2:
                break
            time.sleep(0.3)
        assert len(creates) >= 2, "mid-flight job was not relaunched"
        # and the relaunched gang is actually running under the new manager
        deadline = time.time() + 30
        while time.time() < deadline and not plat2.pytorchjob.gangs:
            time.sleep(0.2)
        assert plat2.pytorchjob.gangs
