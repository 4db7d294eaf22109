"""Reconciler framework — controller-runtime's manager/reconcile loop,
re-built for the in-process object store.

Semantics mirrored from the reference controllers:
  * a watch feeds a de-duplicating work queue of (namespace, name) keys
    (SetupWithManager: notebook_controller.go:573-658);
  * one reconcile at a time per key, level-triggered (the reconciler reads
    current state from the store and converges it);
  * errors or RequeueAfter re-enqueue with delay (culling requeues every
    minute: notebook_controller.go:253-270);
  * ConflictError (stale resourceVersion) is an automatic retry, like k8s
    409s under optimistic concurrency (SURVEY.md §5 race-detection notes).
"""
from __future__ import annotations

import heapq
import threading
import time
import traceback
from typing import Dict, List, Optional, Tuple

from kubeflow_amd.api import ObjectStore, ConflictError, NotFoundError
from kubeflow_amd.api.store import Event


class RequeueAfter(Exception):
    """Raise from reconcile() to re-run after `seconds` (not an error)."""

    def __init__(self, seconds: float):
        self.seconds = seconds
        super().__init__(f"requeue after {seconds}s")


Key = Tuple[Optional[str], str]  # (namespace, name)


class Reconciler:
    """Subclass and implement reconcile(namespace, name)."""

    kind: str = ""                 # the primary kind this controller owns
    watches: List[str] = []        # additional kinds mapped to owner keys

    def __init__(self, store: ObjectStore):
        self.store = store

    def reconcile(self, namespace: Optional[str], name: str) -> None:
        raise NotImplementedError

    def on_deleted(self, namespace: Optional[str], name: str) -> None:
        """Called when the primary object no longer exists (cleanup hook —
        the finalizer analog of profile_controller.go:277-312)."""

    def map_event(self, ev: Event) -> Optional[Key]:
        """Map a watched (non-primary) object event to a primary key — the
        equivalent of the reference's owner-reference/EnqueueRequestsFrom
        mappings (notebook_controller.go:595-658)."""
        for ref in ev.obj["metadata"].get("ownerReferences", []):
            if ref.get("kind") == self.kind:
                return (ev.obj["metadata"].get("namespace"), ref["name"])
        return None


class _Worker(threading.Thread):
    def __init__(self, rec: Reconciler, name: str):
        super().__init__(daemon=True, name=name)
        self.rec = rec
        self.queue: List[Tuple[float, int, Key]] = []  # (ready, seq, key) heap
        self._seq = 0
        self.queued: Dict[Key, float] = {}  # key -> earliest scheduled ready
        self.cv = threading.Condition()
        self.stopping = False
        self.reconcile_count = 0
        self.error_count = 0

    def enqueue(self, key: Key, delay: float = 0.0):
        with self.cv:
            ready = time.monotonic() + delay
            cur = self.queued.get(key)
            if cur is not None and cur <= ready:
                return  # an equal-or-earlier run is already scheduled
            self.queued[key] = ready
            self._seq += 1
            heapq.heappush(self.queue, (ready, self._seq, key))
            self.cv.notify()

    def run(self):
        while True:
            with self.cv:
                while not self.stopping:
                    if self.queue:
                        ready, _, key = self.queue[0]
                        if self.queued.get(key) != ready:
                            heapq.heappop(self.queue)  # superseded entry
                            continue
                        wait = ready - time.monotonic()
                        if wait <= 0:
                            heapq.heappop(self.queue)
                            self.queued.pop(key, None)
                            break
                        self.cv.wait(timeout=min(wait, 1.0))
                    else:
                        self.cv.wait(timeout=1.0)
                if self.stopping:
                    return
            self._process(key)

    def _process(self, key: Key):
        self.reconcile_count += 1
        try:
            self.rec.reconcile(*key)
        except RequeueAfter as rq:
            self.enqueue(key, rq.seconds)
        except ConflictError:
            self.enqueue(key, 0.05)
        except NotFoundError:
            try:
                self.rec.on_deleted(*key)
            except Exception:
                traceback.print_exc()
        except Exception:
            self.error_count += 1
            traceback.print_exc()
            self.enqueue(key, 1.0)

    def stop(self):
        with self.cv:
            self.stopping = True
            self.cv.notify_all()


class ControllerManager:
    """Owns the store watches and one worker thread per reconciler."""

    def __init__(self, store: ObjectStore):
        self.store = store
        self.workers: Dict[str, _Worker] = {}

    def register(self, rec: Reconciler):
        worker = _Worker(rec, name=f"reconcile-{rec.kind}")
        self.workers[rec.kind] = worker

        def on_primary(ev: Event):
            worker.enqueue((ev.obj["metadata"].get("namespace"),
                            ev.obj["metadata"]["name"]))

        self.store.watch(on_primary, kind=rec.kind)
        for wk in rec.watches:
            def on_secondary(ev: Event, rec=rec, worker=worker):
                key = rec.map_event(ev)
                if key is not None:
                    worker.enqueue(key)
            self.store.watch(on_secondary, kind=wk)
        return rec

    def start(self):
        for w in self.workers.values():
            w.start()
        # enqueue all pre-existing objects (informer initial list)
        for kind, w in self.workers.items():
            for obj in self.store.list(kind):
                w.enqueue((obj["metadata"].get("namespace"),
                           obj["metadata"]["name"]))

    def stop(self):
        for w in self.workers.values():
            w.stop()

    def wait_settled(self, timeout: float = 10.0, idle_for: float = 0.2) -> bool:
        """Test helper: wait until all queues have been empty for idle_for."""
        deadline = time.monotonic() + timeout
        settled_since = None
        while time.monotonic() < deadline:
            busy = any(w.queue for w in self.workers.values())
            if busy:
                settled_since = None
            elif settled_since is None:
                settled_since = time.monotonic()
            elif time.monotonic() - settled_since >= idle_for:
                return True
            time.sleep(0.02)
        return False
