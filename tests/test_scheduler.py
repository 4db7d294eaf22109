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
