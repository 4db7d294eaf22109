"""In-process object store — the kube-apiserver + etcd replacement.

Provides, for the single-node platform, what the reference gets from the
Kubernetes apiserver:
  * typed records keyed by (kind, namespace, name),
  * optimistic concurrency via metadata.resourceVersion (update with a stale
    version raises ConflictError, like a k8s 409 — controllers requeue),
  * watch/pub-sub: subscribers get (ADDED/MODIFIED/DELETED, obj) events —
    the informer bus of SURVEY.md §5 collapses into this,
  * optional JSON-lines persistence (durable restart, replay on load),
  * Events (kind="Event") recorded like k8s Events for the activities feed
    (centraldashboard/app/api.ts:66-71 behavior parity).

Thread-safe: one big RLock (single-node control plane; mutation rates are
human/job scale, not data-plane scale).
"""
from __future__ import annotations

import copy
import json
import os
import re
import threading
import time
from typing import Callable, Dict, List, Optional, Tuple

from .objects import KfObject, match_labels, new_object, now_iso


class StoreError(Exception):
    status = 500


class NotFoundError(StoreError):
    status = 404


class AlreadyExistsError(StoreError):
    status = 409


class ConflictError(StoreError):
    status = 409


class Event:
    __slots__ = ("type", "obj")

    def __init__(self, etype: str, obj: KfObject):
        self.type = etype  # ADDED | MODIFIED | DELETED
        self.obj = obj

    def __repr__(self):
        m = self.obj.get("metadata", {})
        return (f"Event({self.type} {self.obj.get('kind')} "
                f"{m.get('namespace')}/{m.get('name')})")


Key = Tuple[str, Optional[str], str]  # (kind, namespace, name)


class InvalidNameError(StoreError):
    status = 422


# Kube name validation (the reference gets this from the kube-apiserver;
# here the store is the apiserver). Names are DNS-1123 subdomains (dots
# allowed — Events use dotted names), namespaces are DNS-1123 labels.
# Without this, a crafted name like '../../x' flows into os.path.join in
# controllers that derive workdirs/log paths from metadata.
_DNS1123_LABEL = re.compile(r"^[a-z0-9]([-a-z0-9]*[a-z0-9])?$")
_DNS1123_SUBDOMAIN = re.compile(
    r"^[a-z0-9]([-a-z0-9]*[a-z0-9])?(\.[a-z0-9]([-a-z0-9]*[a-z0-9])?)*$")
# RBAC object names in kube are path-segment validated, not DNS-1123
# (the reference's owner RoleBinding is literally "namespaceAdmin",
# profile_controller.go:223-244).
_PATH_SEGMENT_KINDS = frozenset({"RoleBinding", "Role", "ClusterRole",
                                 "ClusterRoleBinding"})


def validate_metadata(obj: KfObject) -> None:
    m = obj.get("metadata") or {}
    name = m.get("name")
    if obj.get("kind") in _PATH_SEGMENT_KINDS:
        if not isinstance(name, str) or not name or len(name) > 253 or \
                "/" in name or "%" in name or name in (".", ".."):
            raise InvalidNameError(
                f"invalid name {name!r}: must be a valid path segment")
    elif not isinstance(name, str) or len(name) > 253 or \
            not _DNS1123_SUBDOMAIN.match(name):
        raise InvalidNameError(
            f"invalid name {name!r}: must be a DNS-1123 subdomain "
            "([a-z0-9-.], start/end alphanumeric, <=253 chars)")
    ns = m.get("namespace")
    if ns is not None and (not isinstance(ns, str) or len(ns) > 63 or
                           not _DNS1123_LABEL.match(ns)):
        raise InvalidNameError(
            f"invalid namespace {ns!r}: must be a DNS-1123 label "
            "([a-z0-9-], start/end alphanumeric, <=63 chars)")


class ObjectStore:
    def __init__(self, persist_path: Optional[str] = None):
        self._objs: Dict[Key, KfObject] = {}
        self._lock = threading.RLock()
        self._rv = 0
        self._watchers: List[Tuple[Optional[str], Callable[[Event], None]]] = []
        self._persist_path = persist_path
        self._persist_f = None
        self._persist_lines = 0
        if persist_path:
            self._load(persist_path)
            os.makedirs(os.path.dirname(persist_path) or ".", exist_ok=True)
            # the log is append-only (etcd-WAL analog) and every update adds
            # a line — compact to a snapshot of live objects on load, and
            # again whenever the log grows well past the live set
            self._compact_locked()

    # ------------------------------------------------------------ persist
    def _load(self, path: str):
        if not os.path.exists(path):
            return
        with open(path, encoding="utf-8") as f:
            for line in f:
                line = line.strip()
                if not line:
                    continue
                rec = json.loads(line)
                key = (rec["obj"]["kind"],
                       rec["obj"]["metadata"].get("namespace"),
                       rec["obj"]["metadata"]["name"])
                if rec["op"] == "DELETE":
                    self._objs.pop(key, None)
                else:
                    self._objs[key] = rec["obj"]
                    self._rv = max(self._rv,
                                   int(rec["obj"]["metadata"]["resourceVersion"]))

    def _compact_locked(self):
        """Rewrite the log as one PUT per live object (atomic rename).
        Caller must hold _lock (or be in __init__ before threads exist)."""
        path = self._persist_path
        if self._persist_f:
            self._persist_f.close()
        tmp = path + ".tmp"
        with open(tmp, "w", encoding="utf-8") as f:
            for obj in self._objs.values():
                f.write(json.dumps({"op": "PUT", "obj": obj}) + "\n")
        os.replace(tmp, path)
        self._persist_f = open(path, "a", encoding="utf-8")
        self._persist_lines = len(self._objs)

    def _persist(self, op: str, obj: KfObject):
        if self._persist_f:
            self._persist_f.write(json.dumps({"op": op, "obj": obj}) + "\n")
            self._persist_f.flush()
            self._persist_lines += 1
            if self._persist_lines > max(4096, 8 * len(self._objs)):
                self._compact_locked()

    # -------------------------------------------------------------- watch
    def watch(self, callback: Callable[[Event], None],
              kind: Optional[str] = None) -> None:
        """Register a watcher. Called synchronously under dispatch (keep the
        callback cheap — controllers just enqueue a work item)."""
        with self._lock:
            self._watchers.append((kind, callback))

    def _notify(self, etype: str, obj: KfObject):
        ev = Event(etype, copy.deepcopy(obj))
        for kind, cb in list(self._watchers):
            if kind is None or kind == obj["kind"]:
                try:
                    cb(ev)
                except Exception:  # watcher bugs must not poison the store
                    import traceback
                    traceback.print_exc()

    # --------------------------------------------------------------- CRUD
    def create(self, obj: KfObject) -> KfObject:
        validate_metadata(obj)
        with self._lock:
            key = self._key(obj)
            if key in self._objs:
                raise AlreadyExistsError(f"{key} already exists")
            obj = copy.deepcopy(obj)
            self._rv += 1
            obj["metadata"]["resourceVersion"] = str(self._rv)
            obj["metadata"].setdefault("creationTimestamp", now_iso())
            self._objs[key] = obj
            self._persist("PUT", obj)
            out = copy.deepcopy(obj)
        self._notify("ADDED", out)
        return out

    def get(self, kind: str, name: str,
            namespace: Optional[str] = "default") -> KfObject:
        with self._lock:
            key = (kind, namespace, name)
            if key not in self._objs:
                raise NotFoundError(f"{kind} {namespace}/{name} not found")
            return copy.deepcopy(self._objs[key])

    def list(self, kind: str, namespace: Optional[str] = None,
             label_selector: Optional[Dict[str, str]] = None) -> List[KfObject]:
        with self._lock:
            out = []
            for (k, ns, _), obj in self._objs.items():
                if k != kind:
                    continue
                if namespace is not None and ns != namespace:
                    continue
                if not match_labels(obj, label_selector):
                    continue
                out.append(copy.deepcopy(obj))
            out.sort(key=lambda o: (o["metadata"].get("namespace") or "",
                                    o["metadata"]["name"]))
            return out

    def update(self, obj: KfObject, check_version: bool = True) -> KfObject:
        with self._lock:
            key = self._key(obj)
            if key not in self._objs:
                raise NotFoundError(f"{key} not found")
            cur = self._objs[key]
            if check_version and (obj["metadata"].get("resourceVersion")
                                  != cur["metadata"]["resourceVersion"]):
                raise ConflictError(
                    f"{key}: stale resourceVersion "
                    f"{obj['metadata'].get('resourceVersion')} != "
                    f"{cur['metadata']['resourceVersion']}")
            # no-op writes neither bump the version nor fire watch events
            # (prevents reconcile storms: update -> event -> reconcile -> ...)
            if _same_except_version(obj, cur):
                return copy.deepcopy(cur)
            obj = copy.deepcopy(obj)
            self._rv += 1
            obj["metadata"]["resourceVersion"] = str(self._rv)
            obj["metadata"]["uid"] = cur["metadata"]["uid"]
            self._objs[key] = obj
            self._persist("PUT", obj)
            out = copy.deepcopy(obj)
        self._notify("MODIFIED", out)
        return out

    def patch(self, kind: str, name: str, namespace: Optional[str],
              patch: dict) -> KfObject:
        """Strategic-merge-ish patch (dict deep-merge; None deletes a key).
        Retries internally on conflict — patch semantics are last-writer-wins
        per field, like the k8s PATCH verb the web apps use
        (crud-web-apps .../patch.py:58-66)."""
        for attempt in range(100):
            cur = self.get(kind, name, namespace)
            merged = _deep_merge(cur, patch)
            try:
                return self.update(merged)
            except ConflictError:
                if attempt > 10:
                    time.sleep(0.001 * (attempt - 10))
                continue
        raise ConflictError(f"patch {kind} {namespace}/{name}: retries exhausted")

    def delete(self, kind: str, name: str,
               namespace: Optional[str] = "default") -> KfObject:
        with self._lock:
            key = (kind, namespace, name)
            if key not in self._objs:
                raise NotFoundError(f"{kind} {namespace}/{name} not found")
            obj = self._objs.pop(key)
            self._persist("DELETE", obj)
            out = copy.deepcopy(obj)
        self._notify("DELETED", out)
        # cascade: delete children owning a reference to this uid
        self._delete_owned(out)
        return out

    def _delete_owned(self, owner: KfObject):
        uid = owner["metadata"]["uid"]
        with self._lock:
            children = [o for o in self._objs.values()
                        if any(r.get("uid") == uid
                               for r in o["metadata"].get("ownerReferences", []))]
        for child in children:
            try:
                self.delete(child["kind"], child["metadata"]["name"],
                            child["metadata"].get("namespace"))
            except NotFoundError:
                pass

    # -------------------------------------------------------------- events
    # kube prunes Events by TTL (kube-apiserver --event-ttl, default 1h);
    # without it a long-running platform accumulates them without bound
    EVENT_TTL_S = float(os.environ.get("KF_EVENT_TTL_S", "3600"))
    _EVENT_PRUNE_EVERY = 200

    def _prune_events(self):
        cutoff = time.time() - self.EVENT_TTL_S
        stale = []
        with self._lock:
            for (kind, ns, name), obj in self._objs.items():
                if kind != "Event":
                    continue
                ts = obj["metadata"].get("creationTimestamp", "")
                try:
                    import calendar
                    t = calendar.timegm(time.strptime(ts,
                                                      "%Y-%m-%dT%H:%M:%SZ"))
                except (ValueError, TypeError):
                    continue
                if t < cutoff:
                    stale.append((name, ns))
        for name, ns in stale:
            try:
                self.delete("Event", name, ns)
            except StoreError:
                pass

    def record_event(self, involved: KfObject, reason: str, message: str,
                     etype: str = "Normal"):
        """k8s-Event-shaped record for the activities feed / status surfacing
        (jupyter status derivation reads these: apps/common/status.py:60-99)."""
        if self._rv % self._EVENT_PRUNE_EVERY == 0:
            self._prune_events()
        name = f"{involved['metadata']['name']}.{self._rv}.{int(time.time()*1000)}"
        ev = new_object("Event", name,
                        involved["metadata"].get("namespace") or "default",
                        api_version="v1")
        ev["involvedObject"] = {
            "kind": involved["kind"],
            "name": involved["metadata"]["name"],
            "namespace": involved["metadata"].get("namespace"),
            "uid": involved["metadata"]["uid"],
        }
        ev["reason"] = reason
        ev["message"] = message
        ev["type"] = etype
        ev["lastTimestamp"] = now_iso()
        try:
            return self.create(ev)
        except AlreadyExistsError:
            return None

    def events_for(self, involved: KfObject) -> List[KfObject]:
        evs = self.list("Event", involved["metadata"].get("namespace"))
        uid = involved["metadata"]["uid"]
        mine = [e for e in evs if e.get("involvedObject", {}).get("uid") == uid]
        mine.sort(key=lambda e: int(e["metadata"]["resourceVersion"]))
        return mine

    @staticmethod
    def _key(obj: KfObject) -> Key:
        return (obj["kind"], obj["metadata"].get("namespace"),
                obj["metadata"]["name"])


def _same_except_version(a: KfObject, b: KfObject) -> bool:
    ma, mb = a.get("metadata", {}), b.get("metadata", {})
    if {k: v for k, v in ma.items() if k != "resourceVersion"} != \
            {k: v for k, v in mb.items() if k != "resourceVersion"}:
        return False
    return {k: v for k, v in a.items() if k != "metadata"} == \
        {k: v for k, v in b.items() if k != "metadata"}


def _deep_merge(base: dict, patch: dict) -> dict:
    out = copy.deepcopy(base)
    stack = [(out, patch)]
    while stack:
        dst, src = stack.pop()
        for k, v in src.items():
            if v is None:
                dst.pop(k, None)
            elif isinstance(v, dict) and isinstance(dst.get(k), dict):
                stack.append((dst[k], v))
            else:
                dst[k] = copy.deepcopy(v)
    return out
