"""CRD-shaped object model: {apiVersion, kind, metadata, spec, status}.

Mirrors the conventions of the reference's CRDs — metadata/spec/status with
a conditions[] list (notebook-controller/api/v1beta1/notebook_types.go:27-71,
profile-controller/api/v1/profile_types.go:39-48) — without Kubernetes.
Objects are plain dicts (JSON-serializable end-to-end); this module provides
constructors and helpers, not classes, so REST payloads pass through
unchanged like the k8s custom-objects API does.
"""
from __future__ import annotations

import time
import uuid
from typing import Any, Dict, List, Optional

KfObject = Dict[str, Any]
Condition = Dict[str, Any]


def now_iso() -> str:
    return time.strftime("%Y-%m-%dT%H:%M:%SZ", time.gmtime())


def new_object(kind: str, name: str, namespace: Optional[str] = "default",
               spec: Optional[dict] = None, api_version: str = "kubeflow.org/v1",
               labels: Optional[dict] = None,
               annotations: Optional[dict] = None) -> KfObject:
    meta = {
        "name": name,
        "uid": str(uuid.uuid4()),
        "resourceVersion": "0",
        "creationTimestamp": now_iso(),
        "labels": labels or {},
        "annotations": annotations or {},
        "ownerReferences": [],
    }
    if namespace is not None:
        meta["namespace"] = namespace
    return {
        "apiVersion": api_version,
        "kind": kind,
        "metadata": meta,
        "spec": spec or {},
        "status": {"conditions": []},
    }


def set_condition(obj: KfObject, ctype: str, status: str = "True",
                  reason: str = "", message: str = "") -> None:
    """Update-or-append a condition; bumps lastTransitionTime on change.

    Same shape as the reference's JobCondition/NotebookCondition:
    {type, status, reason, message, lastTransitionTime, lastUpdateTime}.
    """
    conds: List[Condition] = obj.setdefault("status", {}).setdefault(
        "conditions", [])
    ts = now_iso()
    for c in conds:
        if c["type"] == ctype:
            if c.get("status") != status:
                c["lastTransitionTime"] = ts
            c.update(status=status, reason=reason, message=message,
                     lastUpdateTime=ts)
            return
    conds.append({"type": ctype, "status": status, "reason": reason,
                  "message": message, "lastTransitionTime": ts,
                  "lastUpdateTime": ts})


def get_condition(obj: KfObject, ctype: str) -> Optional[Condition]:
    for c in obj.get("status", {}).get("conditions", []):
        if c["type"] == ctype:
            return c
    return None


def has_condition(obj: KfObject, ctype: str, status: str = "True") -> bool:
    c = get_condition(obj, ctype)
    return c is not None and c.get("status") == status


def owner_ref(owner: KfObject) -> dict:
    return {
        "apiVersion": owner["apiVersion"],
        "kind": owner["kind"],
        "name": owner["metadata"]["name"],
        "uid": owner["metadata"]["uid"],
    }


def match_labels(obj: KfObject, selector: Optional[Dict[str, str]]) -> bool:
    if not selector:
        return True
    labels = obj.get("metadata", {}).get("labels", {})
    return all(labels.get(k) == v for k, v in selector.items())
