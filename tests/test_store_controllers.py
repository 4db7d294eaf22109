"""Object store + reconciler framework tests (tier-1/tier-2 of SURVEY §4:
fake-client unit tests + envtest-style store+reconciler harness)."""
import threading
import time

import pytest

from kubeflow_amd.api import (ObjectStore, ConflictError, NotFoundError,
                              AlreadyExistsError, new_object, set_condition,
                              get_condition)
from kubeflow_amd.api.objects import owner_ref, has_condition
from kubeflow_amd.controllers.base import (ControllerManager, Reconciler,
                                           RequeueAfter)


def test_store_crud_and_versioning():
    s = ObjectStore()
    nb = new_object("Notebook", "nb1", "alice", spec={"image": "x"})
    created = s.create(nb)
    assert created["metadata"]["resourceVersion"] == "1"
    with pytest.raises(AlreadyExistsError):
        s.create(nb)
    got = s.get("Notebook", "nb1", "alice")
    assert got["spec"]["image"] == "x"
    got["spec"]["image"] = "y"
    s.update(got)
    # stale update conflicts
    with pytest.raises(ConflictError):
        s.update(got)
    assert s.get("Notebook", "nb1", "alice")["spec"]["image"] == "y"
    s.delete("Notebook", "nb1", "alice")
    with pytest.raises(NotFoundError):
        s.get("Notebook", "nb1", "alice")


def test_store_patch_merge():
    s = ObjectStore()
    s.create(new_object("Notebook", "nb", "ns",
                        annotations={"a": "1"}, spec={"cpu": 1, "gpu": 0}))
    s.patch("Notebook", "nb", "ns",
            {"metadata": {"annotations": {"kubeflow-resource-stopped": "now"}},
             "spec": {"gpu": 2}})
    got = s.get("Notebook", "nb", "ns")
    assert got["metadata"]["annotations"] == {
        "a": "1", "kubeflow-resource-stopped": "now"}
    assert got["spec"] == {"cpu": 1, "gpu": 2}
    # None deletes
    s.patch("Notebook", "nb", "ns",
            {"metadata": {"annotations": {"kubeflow-resource-stopped": None}}})
    assert "kubeflow-resource-stopped" not in \
        s.get("Notebook", "nb", "ns")["metadata"]["annotations"]


def test_store_watch_and_label_select():
    s = ObjectStore()
    events = []
    s.watch(lambda e: events.append((e.type, e.obj["metadata"]["name"])),
            kind="Job")
    s.create(new_object("Job", "j1", "ns", labels={"app": "a"}))
    s.create(new_object("Job", "j2", "ns", labels={"app": "b"}))
    s.create(new_object("Other", "x", "ns"))
    assert ("ADDED", "j1") in events and ("ADDED", "j2") in events
    assert all(n != "x" for _, n in events)
    assert [o["metadata"]["name"]
            for o in s.list("Job", "ns", {"app": "a"})] == ["j1"]


def test_store_owner_cascade_delete():
    s = ObjectStore()
    parent = s.create(new_object("Experiment", "e1", "ns"))
    child = new_object("Trial", "t1", "ns")
    child["metadata"]["ownerReferences"] = [owner_ref(parent)]
    s.create(child)
    s.delete("Experiment", "e1", "ns")
    with pytest.raises(NotFoundError):
        s.get("Trial", "t1", "ns")


def test_store_persistence(tmp_path):
    path = str(tmp_path / "store.jsonl")
    s = ObjectStore(persist_path=path)
    s.create(new_object("Profile", "alice", None, spec={"owner": "a@b.c"}))
    s.create(new_object("Profile", "bob", None))
    s.delete("Profile", "bob", None)
    s2 = ObjectStore(persist_path=path)
    assert [p["metadata"]["name"] for p in s2.list("Profile")] == ["alice"]


def test_conditions_transition_time():
    obj = new_object("Job", "j", "ns")
    set_condition(obj, "Running", "True", "r1")
    t1 = get_condition(obj, "Running")["lastTransitionTime"]
    set_condition(obj, "Running", "True", "r2")
    assert get_condition(obj, "Running")["lastTransitionTime"] == t1
    assert get_condition(obj, "Running")["reason"] == "r2"


class _CounterReconciler(Reconciler):
    kind = "Widget"
    watches = ["Gadget"]

    def __init__(self, store):
        super().__init__(store)
        self.seen = []

    def reconcile(self, namespace, name):
        self.seen.append((namespace, name))
        obj = self.store.get("Widget", name, namespace)
        if not has_condition(obj, "Ready"):
            set_condition(obj, "Ready", "True", "Reconciled")
            self.store.update(obj, check_version=False)


def test_reconciler_loop_and_owner_mapping():
    s = ObjectStore()
    mgr = ControllerManager(s)
    rec = _CounterReconciler(s)
    mgr.register(rec)
    mgr.start()
    try:
        w = s.create(new_object("Widget", "w1", "ns"))
        assert mgr.wait_settled(5)
        assert has_condition(s.get("Widget", "w1", "ns"), "Ready")
        n_before = len(rec.seen)
        # owned Gadget event maps to the Widget key
        g = new_object("Gadget", "g1", "ns")
        g["metadata"]["ownerReferences"] = [owner_ref(w)]
        s.create(g)
        assert mgr.wait_settled(5)
        assert len(rec.seen) > n_before
        assert rec.seen[-1] == ("ns", "w1")
    finally:
        mgr.stop()


class _RequeueReconciler(Reconciler):
    kind = "Poller"

    def __init__(self, store):
        super().__init__(store)
        self.count = 0

    def reconcile(self, namespace, name):
        self.count += 1
        if self.count < 3:
            raise RequeueAfter(0.05)


def test_requeue_after():
    s = ObjectStore()
    mgr = ControllerManager(s)
    rec = _RequeueReconciler(s)
    mgr.register(rec)
    mgr.start()
    try:
        s.create(new_object("Poller", "p", "ns"))
        deadline = time.time() + 5
        while rec.count < 3 and time.time() < deadline:
            time.sleep(0.02)
        assert rec.count >= 3
    finally:
        mgr.stop()


def test_store_log_compaction(tmp_path):
    """The JSONL log compacts instead of growing unboundedly (etcd-WAL
    analog); reload after heavy update churn restores the live set."""
    path = str(tmp_path / "store.jsonl")
    s = ObjectStore(persist_path=path)
    for i in range(8):
        s.create(new_object("PyTorchJob", f"j{i}", "default"))
    o = s.get("PyTorchJob", "j0", "default")
    for k in range(6000):  # way past the 4096-line floor
        o["status"]["k"] = k
        s.update(o, check_version=False)
    with open(path) as f:
        lines = sum(1 for _ in f)
    assert lines < 6000, lines  # compaction happened
    s2 = ObjectStore(persist_path=path)
    assert len(s2.list("PyTorchJob", "default")) == 8
    assert s2.get("PyTorchJob", "j0", "default")["status"]["k"] == 5999
    # compact-on-load leaves exactly the live set
    with open(path) as f:
        assert sum(1 for _ in f) == 8


def test_profile_plugin_seam(tmp_path):
    """Plugin interface parity (profile_controller.go:78-84 + the cloud
    IAM plugins' SA-annotation seam): the local WorkloadIdentity plugin
    annotates default-editor and materializes a credential; unknown kinds
    fail the profile loudly; deletion revokes."""
    import json as _json
    import os as _os
    import time as _time

    from kubeflow_amd.platform import Platform

    with Platform(root_dir=str(tmp_path)) as plat:
        prof = new_object("Profile", "team-wi", None,
                          spec={"owner": {"kind": "User",
                                          "name": "a@b.c"},
                                "plugins": [{"kind": "WorkloadIdentity",
                                             "spec": {"identity":
                                                      "svc-team@local"}}]},
                          api_version="kubeflow.org/v1")
        plat.store.create(prof)
        deadline = _time.time() + 30
        while _time.time() < deadline:
            obj = plat.store.get("Profile", "team-wi", None)
            if has_condition(obj, "Ready"):
                break
            _time.sleep(0.2)
        assert has_condition(obj, "Ready"), obj["status"]
        sa = plat.store.get("ServiceAccount", "default-editor", "team-wi")
        assert sa["metadata"]["annotations"][
            "iam.kubeflow.org/local-identity"] == "svc-team@local"
        cred = obj["status"]["plugins"]["WorkloadIdentity"]["credentialPath"]
        assert _json.load(open(cred))["identity"] == "svc-team@local"

        # unknown plugin kind -> Ready False with PluginFailed
        bad = new_object("Profile", "team-bad", None,
                         spec={"owner": {"kind": "User", "name": "a@b.c"},
                               "plugins": [{"kind": "GcpOnlyThing"}]},
                         api_version="kubeflow.org/v1")
        plat.store.create(bad)
        deadline = _time.time() + 30
        while _time.time() < deadline:
            objb = plat.store.get("Profile", "team-bad", None)
            conds = {c["type"]: c for c in objb["status"]["conditions"]}
            if "Ready" in conds and conds["Ready"]["status"] == "False":
                break
            _time.sleep(0.2)
        conds = {c["type"]: c for c in objb["status"]["conditions"]}
        assert conds["Ready"]["status"] == "False"
        assert "GcpOnlyThing" in conds["Ready"]["message"]

        # deletion revokes the credential
        plat.store.delete("Profile", "team-wi", None)
        deadline = _time.time() + 20
        while _time.time() < deadline and _os.path.exists(cred):
            _time.sleep(0.2)
        assert not _os.path.exists(cred)


def test_event_ttl_pruning(monkeypatch):
    """Events expire after KF_EVENT_TTL_S like kube's --event-ttl."""
    store = ObjectStore()
    store.EVENT_TTL_S = 0.2
    obj = store.create(new_object("PyTorchJob", "evt-job", "default"))
    store.record_event(obj, "Old", "stale soon")
    time.sleep(0.4)
    store._prune_events()
    store.record_event(obj, "New", "fresh")
    reasons = [e["reason"] for e in store.events_for(obj)]
    assert reasons == ["New"]
