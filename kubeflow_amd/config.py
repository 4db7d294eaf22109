"""Platform configuration — one YAML tree + env overrides.

Replaces the reference's mix of Go flags, env vars and ConfigMap-mounted
admin YAMLs (SURVEY.md §5 config/flag system):
  * defaults below mirror the reference's knobs (culler env names kept
    verbatim: ENABLE_CULLING / IDLE_TIME / CULLING_CHECK_PERIOD —
    culler.go:24-27; USERID_HEADER etc. — crud_backend/settings.py:3-6);
  * a YAML file (KF_CONFIG env or /etc/kubeflow-amd/config.yaml) deep-merges
    over the defaults;
  * environment variables override single keys (dots become underscores,
    KF_ prefix): KF_SCHEDULER_BUCKET_MB=128 -> scheduler.bucket_mb;
  * the notebook spawner section keeps the admin `value`/`readOnly`
    semantics of spawner_ui_config.yaml (jupyter/apps/common/utils.py:18-45,
    enforced like form.py:16-48): readOnly fields cannot be overridden by
    user requests.
"""
from __future__ import annotations

import copy
import os
from typing import Any, Dict, Optional

import yaml

DEFAULTS: Dict[str, Any] = {
    "api": {
        "port": 8099,
        "userid_header": "kubeflow-userid",
        "userid_prefix": "",
        "disable_auth": False,
        "csrf": False,
        "cluster_admins": [],
    },
    "culler": {
        "enabled": False,           # ENABLE_CULLING
        "idle_minutes": 1440,       # IDLE_TIME
        "check_period_minutes": 1,  # CULLING_CHECK_PERIOD
    },
    "scheduler": {
        "bucket_mb": 64,            # DDP gradient bucket (xGMI tuning)
        "gang_timeout_s": 600,
    },
    "serving": {
        "max_batch": 16,
        "max_slots": 32,
        "max_seq_len": 4096,
    },
    "spawner": {
        # admin-controlled notebook spawner form config (value/readOnly)
        "image": {"value": "kubeflow-amd/session:latest", "readOnly": False},
        "cpu": {"value": "2", "readOnly": False},
        "memory": {"value": "4Gi", "readOnly": False},
        "gpus": {"value": {"num": "none",
                           "vendors": [{"limitsKey": "amd.com/gpu",
                                        "uiName": "AMD"}]},
                 "readOnly": False},
        "shm": {"value": True, "readOnly": True},
    },
}


def _deep_merge(base: dict, over: dict) -> dict:
    out = copy.deepcopy(base)
    for k, v in over.items():
        if isinstance(v, dict) and isinstance(out.get(k), dict):
            out[k] = _deep_merge(out[k], v)
        else:
            out[k] = copy.deepcopy(v)
    return out


def _apply_env(cfg: dict, prefix: str = "KF_") -> None:
    """KF_SECTION_KEY=value overrides cfg[section][key] (best-effort types)."""
    for name, val in os.environ.items():
        if not name.startswith(prefix):
            continue
        path = name[len(prefix):].lower().split("_")
        node = cfg
        # greedy longest-match walk: joined keys may be dict nodes or leaves
        while len(path) > 1:
            matched = False
            for take in range(len(path), 0, -1):
                key = "_".join(path[:take])
                if key in node:
                    if isinstance(node[key], dict) and take < len(path):
                        node = node[key]
                        path = path[take:]
                        matched = True
                        break
                    if not isinstance(node[key], dict) and take == len(path):
                        path = [key]
                        matched = True
                        break
            if not matched:
                break
        if len(path) == 1 and path[0] in node and not isinstance(
                node[path[0]], dict):
            cur = node[path[0]]
            if isinstance(cur, bool):
                node[path[0]] = val.lower() in ("1", "true", "yes")
            elif isinstance(cur, int):
                try:
                    node[path[0]] = int(val)
                except ValueError:
                    pass
            elif isinstance(cur, float):
                node[path[0]] = float(val)
            else:
                node[path[0]] = val


_cached: Optional[dict] = None


def load(path: Optional[str] = None, use_env: bool = True,
         refresh: bool = False) -> dict:
    global _cached
    if _cached is not None and not refresh and path is None:
        return _cached
    cfg = copy.deepcopy(DEFAULTS)
    path = path or os.environ.get("KF_CONFIG",
                                  "/etc/kubeflow-amd/config.yaml")
    if path and os.path.exists(path):
        with open(path) as f:
            over = yaml.safe_load(f) or {}
        cfg = _deep_merge(cfg, over)
    if use_env:
        _apply_env(cfg)
    if path is None or not os.path.exists(path):
        pass
    _cached = cfg
    return cfg


def enforce_spawner(cfg: dict, requested: dict) -> dict:
    """Apply value/readOnly semantics: readOnly fields always take the admin
    value; writable fields take the request's value when present
    (form.py:16-48 behavior)."""
    out = {}
    for field, meta in cfg.get("spawner", {}).items():
        if not isinstance(meta, dict) or "value" not in meta:
            continue
        if meta.get("readOnly"):
            out[field] = meta["value"]
        else:
            out[field] = requested.get(field, meta["value"])
    return out
