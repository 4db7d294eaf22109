"""Process gang launcher — one worker process per MI355X.

Replaces kubelet + CRI + the admission webhook's pod mutation
("in the rebuild this becomes a synchronous defaults
injection pass inside the launcher"):

  * per-rank env: RANK/WORLD_SIZE/LOCAL_RANK/MASTER_ADDR/MASTER_PORT +
    HIP_VISIBLE_DEVICES pinning one GPU per process (the extension seam for
    future TP/PP strategies);
  * PodDefault injection: env/annotations from matching PodDefault objects
    are merged into the worker env before spawn (same selector + conflict
    semantics as admission-webhook/main.go:69-94,369-441);
  * gang semantics: all ranks spawn together; if any rank dies, the whole
    gang is killed (kill-on-peer-death); children get PR_SET_PDEATHSIG so
    a dead controller never leaks GPU processes;
  * NUMA affinity: worker is bound to its GPU's NUMA node via numactl when
    available (falls back to no pinning).
"""
from __future__ import annotations

import ctypes
import json
import os
import shutil
import signal
import socket
import subprocess
import sys
import threading
import time
from dataclasses import dataclass
from typing import Dict, List, Optional

PR_SET_PDEATHSIG = 1


def _preexec():
    # die with the parent (controller) — no leaked GPU processes
    libc = ctypes.CDLL("libc.so.6", use_errno=True)
    libc.prctl(PR_SET_PDEATHSIG, signal.SIGKILL)
    os.setpgrp()


def free_port() -> int:
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def merge_poddefaults(env: Dict[str, str], labels: Dict[str, str],
                      poddefaults: List[dict]) -> Dict[str, str]:
    """Env-only PodDefault merge for callers that have no volume/toleration
    surface (notebook sessions). `env` must be the EXPLICITLY constructed
    env for the workload — never the inherited os.environ: the webhook
    compares PodDefault env only against pod-spec env
    (admission-webhook/main.go:152-187), so a PodDefault var that happens
    to collide with a controller environment variable must still inject.
    Conflicting values raise, mirroring safeToApplyPodDefaultsOnPod."""
    from kubeflow_amd.scheduler.poddefaults import apply_poddefaults
    merged = apply_poddefaults({"env": dict(env), "labels": dict(labels)},
                               poddefaults, labels=labels)
    return merged["env"]


@dataclass
class RankProc:
    rank: int
    proc: subprocess.Popen
    log_path: str


class ProcessGang:
    """A running gang of rank processes for one job."""

    def __init__(self, job_uid: str, workdir: str):
        self.job_uid = job_uid
        self.workdir = workdir
        self.ranks: List[RankProc] = []
        self._killed = False
        self._lock = threading.Lock()
        self.started_at = time.time()

    def poll(self) -> Optional[str]:
        """None while running; 'Succeeded' when all ranks exited 0;
        'Failed' otherwise. Enforces kill-on-peer-death."""
        with self._lock:
            codes = [r.proc.poll() for r in self.ranks]
            if any(c not in (None, 0) for c in codes):
                self._kill_locked()
                return "Failed"
            if all(c == 0 for c in codes):
                return "Succeeded"
            return None

    def kill(self):
        with self._lock:
            self._kill_locked()

    def _kill_locked(self):
        if self._killed:
            return
        self._killed = True
        for r in self.ranks:
            if r.proc.poll() is None:
                try:
                    os.killpg(r.proc.pid, signal.SIGTERM)
                except ProcessLookupError:
                    pass
        deadline = time.time() + 5
        for r in self.ranks:
            try:
                r.proc.wait(timeout=max(0.1, deadline - time.time()))
            except subprocess.TimeoutExpired:
                try:
                    os.killpg(r.proc.pid, signal.SIGKILL)
                except ProcessLookupError:
                    pass

    def terminate_and_wait(self, timeout: float = 10.0):
        self.kill()
        for r in self.ranks:
            try:
                r.proc.wait(timeout=timeout)
            except subprocess.TimeoutExpired:
                pass


def launch_gang(job_uid: str, workdir: str, spec: dict, gpu_indices: List[int],
                poddefaults: Optional[List[dict]] = None,
                labels: Optional[Dict[str, str]] = None,
                numa_nodes: Optional[Dict[int, int]] = None,
                entry_module: str = "kubeflow_amd.runtime.worker",
                warm_pool=None, pvc_root: Optional[str] = None,
                configmaps: Optional[Dict[str, Dict[str, str]]] = None
                ) -> ProcessGang:
    """Spawn one process per rank. CPU jobs pass gpu_indices=[] and
    spec['world_size'] ranks run on CPU (gloo). With a WarmPool, ranks are
    handed to pre-forked workers (torch already imported) when available.

    PodDefault injection covers the webhook's full merge surface
    (env/envFrom/volumes/volumeMounts/tolerations/labels/annotations —
    scheduler/poddefaults.py); volumes materialize as per-rank symlinks
    under rank-N/mnt plus a KF_VOLUME_MOUNTS env table."""
    from kubeflow_amd.scheduler.poddefaults import (
        apply_poddefaults, materialize_mounts, mounts_env, resolve_env_from)
    os.makedirs(workdir, exist_ok=True)
    world = max(1, len(gpu_indices) or int(spec.get("world_size", 1)))
    port = free_port()
    spec_path = os.path.join(workdir, "spec.json")
    with open(spec_path, "w") as f:
        json.dump(spec, f, indent=2)

    gang = ProcessGang(job_uid, workdir)
    repo_root = os.path.dirname(os.path.dirname(
        os.path.dirname(os.path.abspath(__file__))))
    merged = apply_poddefaults(
        {"env": {k: str(v) for k, v in (spec.get("env") or {}).items()},
         "env_from": spec.get("env_from") or [],
         "volumes": spec.get("volumes") or [],
         "volume_mounts": spec.get("volume_mounts") or [],
         "tolerations": spec.get("tolerations") or [],
         "labels": dict(labels or {})},
        poddefaults or [], labels=labels or {})
    injected = resolve_env_from(merged["env_from"], configmaps or {},
                                env=merged["env"])
    for rank in range(world):
        rank_dir = os.path.join(workdir, f"rank-{rank}")
        os.makedirs(rank_dir, exist_ok=True)
        mounts = materialize_mounts(rank_dir, merged["volumes"],
                                    merged["volume_mounts"],
                                    pvc_root=pvc_root)
        rank_injected = dict(injected)
        rank_injected.update(mounts_env(mounts))
        if warm_pool is not None and entry_module == \
                "kubeflow_amd.runtime.worker":
            got = warm_pool.take()
            if got is not None:
                proc, log_path = got
                warm_pool.assign(proc, {
                    "spec_path": spec_path,
                    "rank": rank,
                    "world_size": world,
                    "master_port": port,
                    "workdir": workdir,
                    "job_uid": job_uid,
                    "gpu": gpu_indices[rank] if gpu_indices else None,
                    "env": rank_injected,
                })
                gang.ranks.append(RankProc(rank, proc, log_path))
                continue
        env = dict(os.environ)
        env.update(rank_injected)  # PodDefault/envFrom merge (explicit-only)
        env.update({
            "RANK": str(rank),
            "WORLD_SIZE": str(world),
            "LOCAL_RANK": str(rank),
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
            "KF_JOB_UID": job_uid,
            "KF_JOB_WORKDIR": workdir,
            "PYTHONPATH": repo_root + os.pathsep +
                          os.environ.get("PYTHONPATH", ""),
        })
        env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        if gpu_indices:
            env["HIP_VISIBLE_DEVICES"] = str(gpu_indices[rank])
            env["LOCAL_RANK"] = "0"  # each proc sees exactly one GPU

        cmd = [sys.executable, "-m", entry_module, "--spec", spec_path]
        if gpu_indices and numa_nodes and shutil.which("numactl"):
            node = numa_nodes.get(gpu_indices[rank], 0)
            cmd = ["numactl", f"--cpunodebind={node}",
                   f"--preferred={node}"] + cmd
        log_path = os.path.join(rank_dir, "worker.log")
        logf = open(log_path, "w")
        proc = subprocess.Popen(cmd, env=env, stdout=logf, stderr=logf,
                                cwd=rank_dir, preexec_fn=_preexec)
        gang.ranks.append(RankProc(rank, proc, log_path))
    return gang
