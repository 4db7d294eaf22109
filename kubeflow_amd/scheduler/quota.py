"""ResourceQuota enforcement at admission.

The reference's profile controller creates a ResourceQuota per namespace
and kube enforces it (profile_controller.go:425-455); here the store holds
the quota objects (controllers/profile.py) and admission happens where
kubelet's admission would — before the gang scheduler grants GPUs.

Recognized hard limits (quota spec {"hard": {...}}): `amd.com/gpu`,
`requests.amd.com/gpu` (GPU count) — checked against the scheduler's live
per-namespace allocations plus the request.
"""
from __future__ import annotations

from typing import Optional

GPU_KEYS = ("requests.amd.com/gpu", "amd.com/gpu", "limits.amd.com/gpu")


class QuotaExceeded(Exception):
    pass


def gpu_limit(store, namespace: Optional[str]) -> Optional[int]:
    """The namespace's GPU quota, or None when unlimited/no quota."""
    if namespace is None:
        return None
    limit = None
    for q in store.list("ResourceQuota", namespace):
        hard = (q.get("spec") or {}).get("hard") or {}
        for key in GPU_KEYS:
            if key in hard:
                v = int(hard[key])
                limit = v if limit is None else min(limit, v)
    return limit


def admit_gpus(store, scheduler, namespace: Optional[str],
               requested: int) -> None:
    """Raise QuotaExceeded if granting `requested` GPUs would push the
    namespace past its ResourceQuota."""
    if requested <= 0:
        return
    limit = gpu_limit(store, namespace)
    if limit is None:
        return
    used = scheduler.ns_gpu_usage(namespace)
    if used + requested > limit:
        raise QuotaExceeded(
            f"namespace {namespace}: ResourceQuota limits amd.com/gpu to "
            f"{limit}; {used} in use, {requested} requested")
