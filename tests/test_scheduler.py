"""Gang scheduler unit tests (fake GPUs)."""
import pytest

from kubeflow_amd.scheduler.gang import GangScheduler, InsufficientResources
from kubeflow_amd.scheduler.inventory import GpuInventory


@pytest.fixture
def sched(monkeypatch):
    monkeypatch.setenv("KF_FAKE_GPUS", "8")
    return GangScheduler(GpuInventory())


def test_all_or_nothing_and_contiguous(sched):
    a = sched.allocate("job1", 4)
    assert a.gpu_indices == [0, 1, 2, 3]
    b = sched.allocate("job2", 4)
    assert b.gpu_indices == [4, 5, 6, 7]
    with pytest.raises(InsufficientResources):
        sched.allocate("job3", 1)
    sched.release("job1")
    c = sched.allocate("job3", 2)
    assert c.gpu_indices == [0, 1]
    assert sched.utilization()["exclusive_busy"] == 6


def test_contiguous_preference_with_hole(sched):
    for i, name in enumerate(["a", "b", "c", "d"]):
        sched.allocate(name, 2)
    sched.release("b")  # hole at 2,3
    e = sched.allocate("e", 2)
    assert e.gpu_indices == [2, 3]  # fills the xGMI-contiguous hole


def test_idempotent_allocate_and_zero_gpu(sched):
    a1 = sched.allocate("j", 3)
    a2 = sched.allocate("j", 3)
    assert a1.gpu_indices == a2.gpu_indices
    z = sched.allocate("cpu-job", 0)
    assert z.gpu_indices == []
    sched.release("cpu-job")
    sched.release("nonexistent")  # no-op


def test_shared_memory_allocation(sched):
    gib = 1024 ** 3
    a = sched.allocate("svc1", 1, mem_per_gpu=100 * gib, exclusive=False)
    b = sched.allocate("svc2", 1, mem_per_gpu=100 * gib, exclusive=False)
    # both fit on GPU 0 (288 GiB)
    assert a.gpu_indices == [0] and b.gpu_indices == [0]
    c = sched.allocate("svc3", 1, mem_per_gpu=100 * gib, exclusive=False)
    assert c.gpu_indices == [1]  # GPU0 would exceed 288 GiB
    # exclusive job skips partially-used GPUs
    d = sched.allocate("train", 6)
    assert 0 not in d.gpu_indices and 1 not in d.gpu_indices


# --------------------------------------------------- ResourceQuota admission

def test_quota_admission(monkeypatch):
    """Namespace ResourceQuota caps GPU grants across notebook + job
    allocations (the reference delegates this to kube enforcing what the
    profile controller creates — profile_controller.go:425-455)."""
    from kubeflow_amd.api import new_object
    from kubeflow_amd.api.store import ObjectStore
    from kubeflow_amd.scheduler.gang import GangScheduler, GpuInventory
    from kubeflow_amd.scheduler.quota import QuotaExceeded, admit_gpus

    monkeypatch.setenv("KF_FAKE_GPUS", "8")
    store = ObjectStore()
    q = new_object("ResourceQuota", "kf-resource-quota", "team-a",
                   api_version="v1",
                   spec={"hard": {"requests.amd.com/gpu": 2}})
    store.create(q)
    sched = GangScheduler(GpuInventory())
    assert sched.inv.n_gpus >= 4

    admit_gpus(store, sched, "team-a", 2)           # within quota
    sched.allocate("job-1", 2, namespace="team-a")
    assert sched.ns_gpu_usage("team-a") == 2
    with pytest.raises(QuotaExceeded):              # N+1th GPU rejected
        admit_gpus(store, sched, "team-a", 1)
    admit_gpus(store, sched, "team-b", 8)           # other ns unlimited
    sched.release("job-1")
    admit_gpus(store, sched, "team-a", 2)           # released -> ok again


def test_quota_blocks_gpu_job_e2e(tmp_path, monkeypatch):
    """e2e: a quota-limited namespace rejects the job that would exceed it
    with a QuotaExceeded Warning event, and the job starts once quota
    frees (VERDICT item 6 done-criterion)."""
    import time as _t

    from kubeflow_amd.api import new_object
    from kubeflow_amd.api.objects import has_condition
    from kubeflow_amd.platform import Platform

    monkeypatch.setenv("KF_FAKE_GPUS", "4")
    with Platform(root_dir=str(tmp_path)) as plat:
        plat.store.create(new_object(
            "ResourceQuota", "kf-resource-quota", "default",
            api_version="v1", spec={"hard": {"requests.amd.com/gpu": 1}}))
        job = new_object("PyTorchJob", "too-big", "default", spec={
            "pytorchReplicaSpecs": {"Worker": {
                "replicas": 2, "restartPolicy": "Never",
                "template": {"model": "mnist-mlp", "steps": 2,
                             "gpus_per_replica": 1}}}})
        plat.store.create(job)
        deadline = _t.time() + 30
        evs = []
        while _t.time() < deadline:
            obj = plat.store.get("PyTorchJob", "too-big", "default")
            evs = [e["reason"] for e in plat.store.events_for(obj)]
            if "QuotaExceeded" in evs:
                break
            _t.sleep(0.3)
        assert "QuotaExceeded" in evs, evs
        assert not has_condition(obj, "Running")
        # raising the quota lets it through
        q = plat.store.get("ResourceQuota", "kf-resource-quota", "default")
        q["spec"]["hard"]["requests.amd.com/gpu"] = 2
        plat.store.update(q, check_version=False)
        deadline = _t.time() + 60
        while _t.time() < deadline:
            obj = plat.store.get("PyTorchJob", "too-big", "default")
            if has_condition(obj, "Running") or \
                    has_condition(obj, "Succeeded") or \
                    has_condition(obj, "Failed"):
                break
            _t.sleep(0.3)
        assert has_condition(obj, "Running") or \
            has_condition(obj, "Succeeded") or has_condition(obj, "Failed")
