"""Warm worker pool — pre-forked ranks for fast gang start.

Keeps N `kubeflow_amd.runtime.warm_worker` processes alive with torch
already imported (the dominant cold-start cost); `launch_gang` takes workers
from the pool and hands each its rank assignment over stdin. The pool
replenishes itself in the background after every take.

Trade-offs (documented, deliberate):
  * warm workers are not numactl-pinned (they pre-exist their GPU
    assignment); HBM/NUMA locality still holds because HIP_VISIBLE_DEVICES
    is applied before the worker's first CUDA call;
  * worker stdout/stderr go to the pool's log files (status/error state
    still flows through the rank status.json protocol).
"""
from __future__ import annotations

import json
import os
import subprocess
import sys
import threading
import time
from typing import List, Optional, Tuple

from .launcher import _preexec


class WarmPool:
    def __init__(self, size: int, pool_dir: str):
        self.size = size
        self.pool_dir = pool_dir
        os.makedirs(pool_dir, exist_ok=True)
        self._lock = threading.Lock()
        self._idle: List[Tuple[subprocess.Popen, str]] = []
        self._seq = 0
        self._closed = False
        # ALL spawns happen on this one persistent thread: PR_SET_PDEATHSIG
        # kills the child when its spawning THREAD exits, so spawning from
        # short-lived threads would SIGKILL warm workers immediately (and
        # this thread dying with the platform is exactly the cleanup we
        # want).
        self._wake = threading.Event()
        self._spawner = threading.Thread(target=self._spawn_loop, daemon=True,
                                         name="warmpool-spawner")
        self._spawner.start()
        self._wake.set()

    def _spawn_loop(self):
        while True:
            self._wake.wait()
            self._wake.clear()
            if self._closed:
                return
            while True:
                with self._lock:
                    need = (not self._closed
                            and len(self._idle) < self.size)
                if not need:
                    break
                try:
                    self._spawn()
                except Exception:
                    # a spawn failure (e.g. transient fork/OOM) must not
                    # kill this thread: its death SIGKILLs every warm
                    # worker via their PDEATHSIG
                    import traceback
                    traceback.print_exc()
                    time.sleep(1.0)

    def _spawn(self):
        with self._lock:
            if self._closed or len(self._idle) >= self.size:
                return
            self._seq += 1
            seq = self._seq
        repo_root = os.path.dirname(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))))
        env = dict(os.environ)
        env["PYTHONPATH"] = repo_root + os.pathsep + env.get("PYTHONPATH", "")
        env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
        log_path = os.path.join(self.pool_dir, f"warm-{seq}.log")
        logf = open(log_path, "w")
        proc = subprocess.Popen(
            [sys.executable, "-m", "kubeflow_amd.runtime.warm_worker"],
            stdin=subprocess.PIPE, stdout=logf, stderr=logf, env=env,
            preexec_fn=_preexec)
        with self._lock:
            if self._closed:
                proc.kill()
                return
            self._idle.append((proc, log_path))

    @staticmethod
    def _ready(log_path: str) -> bool:
        try:
            with open(log_path) as f:
                return "[warm-worker] ready" in f.read()
        except OSError:
            return False

    def take(self) -> Optional[Tuple[subprocess.Popen, str]]:
        """Pop a live, READY warm worker (or None — callers then cold-spawn;
        a worker still paying the torch import is slower than a cold start
        plus it would serialize the handoff). Replenishes in the background."""
        got = None
        with self._lock:
            keep = []
            while self._idle:
                proc, log = self._idle.pop()
                if proc.poll() is not None:
                    continue  # died
                if got is None and self._ready(log):
                    got = (proc, log)
                else:
                    keep.append((proc, log))
            self._idle.extend(keep)
        if got is not None:
            self._wake.set()  # replenish on the persistent spawner thread
        return got

    def wait_ready(self, n: int = 1, timeout: float = 30.0) -> bool:
        deadline = time.time() + timeout
        while time.time() < deadline:
            with self._lock:
                ready = sum(1 for p, log in self._idle
                            if p.poll() is None and self._ready(log))
            if ready >= n:
                return True
            time.sleep(0.1)
        return False

    @staticmethod
    def assign(proc: subprocess.Popen, assignment: dict):
        proc.stdin.write((json.dumps(assignment) + "\n").encode())
        proc.stdin.flush()

    def shutdown(self):
        with self._lock:
            self._closed = True
            idle, self._idle = self._idle, []
        self._wake.set()
        for proc, _ in idle:
            try:
                proc.stdin.close()
                proc.terminate()
            except Exception:
                pass
