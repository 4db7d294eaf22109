"""Gang scheduler — all-or-nothing GPU allocation with bin-packing.

Replaces the K8s scheduler + the gang semantics Kubeflow's training
operators rely on: a job either gets ALL its GPUs or
none (no partial deadlock-prone allocations), and GPU sets are chosen
contiguous-first so RCCL rings run over neighboring xGMI links.

Thread-safe; memory is tracked per-GPU so several 1-GPU jobs (Katib trials,
InferenceServices) can share the node with a big training job.
"""
from __future__ import annotations

import threading
from dataclasses import dataclass
from typing import Dict, List, Optional

from .inventory import GpuInventory


class InsufficientResources(Exception):
    pass


@dataclass
class Allocation:
    job_uid: str
    gpu_indices: List[int]
    mem_per_gpu: int


class GangScheduler:
    def __init__(self, inventory: Optional[GpuInventory] = None):
        self.inv = inventory or GpuInventory()
        self._lock = threading.Lock()
        self._alloc: Dict[str, Allocation] = {}           # job_uid -> alloc
        self._gpu_mem_used: Dict[int, int] = {g.index: 0 for g in self.inv.gpus}
        self._gpu_excl: Dict[int, Optional[str]] = {
            g.index: None for g in self.inv.gpus}
        self._ns_of: Dict[str, str] = {}  # job_uid -> namespace (quota)

    def allocate(self, job_uid: str, n_gpus: int, mem_per_gpu: int = 0,
                 exclusive: bool = True,
                 namespace: Optional[str] = None) -> Allocation:
        """All-or-nothing allocation of n_gpus. exclusive=True (training)
        claims whole GPUs; exclusive=False co-schedules by HBM bytes.
        `namespace` tags the allocation for ResourceQuota accounting."""
        with self._lock:
            if job_uid in self._alloc:
                return self._alloc[job_uid]
            if n_gpus == 0:
                alloc = Allocation(job_uid, [], 0)
                self._alloc[job_uid] = alloc
                return alloc
            free = []
            for g in self.inv.gpus:
                if self._gpu_excl[g.index] is not None:
                    continue
                if exclusive and self._gpu_mem_used[g.index] > 0:
                    continue
                if not exclusive and (self._gpu_mem_used[g.index] + mem_per_gpu
                                      > g.hbm_bytes):
                    continue
                free.append(g.index)
            if len(free) < n_gpus:
                raise InsufficientResources(
                    f"need {n_gpus} GPUs, {len(free)} available "
                    f"(total {self.inv.n_gpus})")
            chosen = self._pick_contiguous(free, n_gpus)
            for idx in chosen:
                if exclusive:
                    self._gpu_excl[idx] = job_uid
                self._gpu_mem_used[idx] += mem_per_gpu
            alloc = Allocation(job_uid, chosen, mem_per_gpu)
            self._alloc[job_uid] = alloc
            if namespace:
                self._ns_of[job_uid] = namespace
            return alloc

    @staticmethod
    def _pick_contiguous(free: List[int], n: int) -> List[int]:
        """Prefer a contiguous run of GPU indices (neighboring xGMI links);
        fall back to the first n free."""
        free = sorted(free)
        for i in range(len(free) - n + 1):
            window = free[i:i + n]
            if window[-1] - window[0] == n - 1:
                return window
        return free[:n]

    def ns_gpu_usage(self, namespace: str) -> int:
        """GPUs currently allocated to a namespace (quota accounting)."""
        with self._lock:
            return sum(len(a.gpu_indices) for uid, a in self._alloc.items()
                       if self._ns_of.get(uid) == namespace)

    def release(self, job_uid: str):
        with self._lock:
            self._ns_of.pop(job_uid, None)
            alloc = self._alloc.pop(job_uid, None)
            if alloc is None:
                return
            for idx in alloc.gpu_indices:
                if self._gpu_excl.get(idx) == job_uid:
                    self._gpu_excl[idx] = None
                self._gpu_mem_used[idx] = max(
                    0, self._gpu_mem_used[idx] - alloc.mem_per_gpu)

    def utilization(self) -> dict:
        with self._lock:
            busy = sum(1 for v in self._gpu_excl.values() if v is not None)
            shared = sum(1 for i, v in self._gpu_mem_used.items()
                         if v > 0 and self._gpu_excl[i] is None)
            return {"total_gpus": self.inv.n_gpus, "exclusive_busy": busy,
                    "shared_busy": shared, "jobs": len(self._alloc)}
