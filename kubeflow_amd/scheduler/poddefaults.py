"""PodDefault merge semantics — the admission-webhook's mutation pass.

In the reference this is a mutating webhook intercepting every pod CREATE
(admission-webhook/main.go:443 mutatePods); in the process model it is a
synchronous pass inside the launcher over a "process spec" that carries the
same fields a pod carries:

    {env: {name: value}, env_from: [...], volumes: [...],
     volume_mounts: [...], tolerations: [...],
     labels: {...}, annotations: {...}}

Behavior parity (each mirrors a reference function):
  * selection: label selector with matchLabels + matchExpressions
    (filterPodDefaults, main.go:68-93; LabelSelectorAsSelector semantics);
  * env: inject if absent; identical re-definition ok; differing value is a
    conflict (mergeEnv, main.go:152-187);
  * envFrom: plain append (mergeEnvFrom, main.go:189-199);
  * volumeMounts: conflict on duplicate name OR duplicate mountPath with
    differing definitions (mergeVolumeMounts, main.go:201-254);
  * volumes: conflict on duplicate name with differing source
    (mergeVolumes, main.go:256-297);
  * tolerations: keyed by `key` (mergeTolerations, main.go:299-341);
  * labels/annotations: add-if-absent, conflict on differing value
    (mergeMap, main.go:343-364);
  * conflicts across ALL fields are collected before raising, mirroring
    safeToApplyPodDefaultsOnPod's error aggregate (main.go:98-132);
  * every applied PodDefault is recorded as an annotation
    poddefault.admission.kubeflow.org/poddefault-<name>=<resourceVersion>
    (applyPodDefaultsOnPod, main.go:418-421).

Volume/volumeMount materialization maps to the PVC-directory model
(controllers/volume.py: a PVC is a directory): the launcher symlinks each
mounted volume under <rank_dir>/mnt/<volume-name> and exports the mount
table as KF_VOLUME_MOUNTS (JSON [{"name","mountPath","hostPath"}...]) so
workloads can resolve the declared mountPath.
"""
from __future__ import annotations

import json
import os
from typing import Dict, List, Optional


class PodDefaultConflict(ValueError):
    """Aggregate of merge conflicts (safeToApplyPodDefaultsOnPod)."""

    def __init__(self, errors: List[str]):
        self.errors = errors
        super().__init__("; ".join(errors))


def selector_matches(selector: dict, labels: Dict[str, str]) -> bool:
    """LabelSelectorAsSelector semantics: matchLabels AND matchExpressions
    (In/NotIn/Exists/DoesNotExist) must all hold."""
    for k, v in (selector.get("matchLabels") or {}).items():
        if labels.get(k) != v:
            return False
    for expr in selector.get("matchExpressions") or []:
        key, op = expr.get("key"), expr.get("operator")
        vals = expr.get("values") or []
        present = key in labels
        if op == "In":
            if not present or labels[key] not in vals:
                return False
        elif op == "NotIn":
            if present and labels[key] in vals:
                return False
        elif op == "Exists":
            if not present:
                return False
        elif op == "DoesNotExist":
            if present:
                return False
        else:
            return False  # unknown operator: fail closed, like a bad selector
    return True


def filter_poddefaults(poddefaults: List[dict], labels: Dict[str, str],
                       namespace: Optional[str] = None) -> List[dict]:
    """filterPodDefaults (main.go:68-93): selector match + same namespace."""
    out = []
    for pd in poddefaults or []:
        if namespace is not None:
            pd_ns = (pd.get("metadata") or {}).get("namespace")
            if pd_ns is not None and pd_ns != namespace:
                continue
        sel = (pd.get("spec") or {}).get("selector") or {}
        if selector_matches(sel, labels or {}):
            out.append(pd)
    return out


def _merge_keyed(existing: List[dict], pds: List[dict], field: str,
                 key: str, errs: List[str], what: str,
                 extra_key: Optional[str] = None) -> List[dict]:
    """Shared add-if-absent / identical-ok / differing-conflict merge used
    for env-list, volumes, volumeMounts and tolerations."""
    orig = {e.get(key): e for e in existing}
    by_extra = {e.get(extra_key): e for e in existing} if extra_key else {}
    merged = list(existing)
    for pd in pds:
        name = (pd.get("metadata") or {}).get("name", "?")
        for item in (pd.get("spec") or {}).get(field) or []:
            k = item.get(key)
            found = orig.get(k)
            if found is None:
                orig[k] = item
                merged.append(item)
                if extra_key is not None:
                    ek = item.get(extra_key)
                    efound = by_extra.get(ek)
                    if efound is None:
                        by_extra[ek] = item
                    elif efound != item:
                        errs.append(
                            f"merging {what} for {name} has a conflict on "
                            f"{extra_key} {ek}")
                continue
            if found != item:
                errs.append(
                    f"merging {what} for {name} has a conflict on {k}")
    return merged


def _merge_map(existing: Dict[str, str], pds: List[dict], field: str,
               errs: List[str]) -> Dict[str, str]:
    out = dict(existing or {})
    for pd in pds:
        for k, v in ((pd.get("spec") or {}).get(field) or {}).items():
            if k not in out:
                out[k] = v
            elif out[k] != v:
                errs.append(f"merging has conflict on {k}")
    return out


def apply_poddefaults(proc_spec: dict, poddefaults: List[dict],
                      labels: Optional[Dict[str, str]] = None,
                      namespace: Optional[str] = None) -> dict:
    """Merge matching PodDefaults into a process spec; raises
    PodDefaultConflict with every conflict if any merge is unsafe
    (the safeToApply + apply pair, collapsed: we never mutate on error).

    `proc_spec` fields are all optional; returns a new dict with
    env / env_from / volumes / volume_mounts / tolerations / labels /
    annotations fully merged.
    """
    if labels is None:
        labels = proc_spec.get("labels") or {}
    pds = filter_poddefaults(poddefaults, labels, namespace)
    errs: List[str] = []

    env_list = [{"name": k, "value": v}
                for k, v in (proc_spec.get("env") or {}).items()]
    merged_env = _merge_keyed(env_list, pds, "env", "name", errs, "env")
    env_from = list(proc_spec.get("env_from") or [])
    for pd in pds:  # mergeEnvFrom: plain append
        env_from.extend((pd.get("spec") or {}).get("envFrom") or [])
    volumes = _merge_keyed(list(proc_spec.get("volumes") or []), pds,
                           "volumes", "name", errs, "volumes")
    mounts = _merge_keyed(list(proc_spec.get("volume_mounts") or []), pds,
                          "volumeMounts", "name", errs, "volume mounts",
                          extra_key="mountPath")
    tolerations = _merge_keyed(list(proc_spec.get("tolerations") or []), pds,
                               "tolerations", "key", errs, "tolerations")
    out_labels = _merge_map(proc_spec.get("labels") or {}, pds, "labels",
                            errs)
    annotations = _merge_map(proc_spec.get("annotations") or {}, pds,
                             "annotations", errs)
    if errs:
        raise PodDefaultConflict(errs)

    for pd in pds:  # mutation marker annotations (main.go:418-421)
        m = pd.get("metadata") or {}
        annotations[
            "poddefault.admission.kubeflow.org/poddefault-"
            + m.get("name", "?")] = str(m.get("resourceVersion", ""))

    return {
        "env": {e["name"]: str(e.get("value", "")) for e in merged_env},
        "env_from": env_from,
        "volumes": volumes,
        "volume_mounts": mounts,
        "tolerations": tolerations,
        "labels": out_labels,
        "annotations": annotations,
    }


def resolve_env_from(env_from: List[dict],
                     configmaps: Dict[str, Dict[str, str]],
                     secrets: Optional[Dict[str, Dict[str, str]]] = None,
                     env: Optional[Dict[str, str]] = None) -> Dict[str, str]:
    """Expand envFrom sources against store ConfigMap/Secret data.
    Explicit env wins over envFrom (kube's container env precedence)."""
    out: Dict[str, str] = {}
    for src in env_from or []:
        prefix = src.get("prefix", "")
        ref = src.get("configMapRef") or {}
        data = configmaps.get(ref.get("name", ""), {}) if ref else {}
        sref = src.get("secretRef") or {}
        if sref and secrets:
            data = secrets.get(sref.get("name", ""), {})
        for k, v in data.items():
            out[prefix + k] = str(v)
    for k, v in (env or {}).items():
        out[k] = v
    return out


def materialize_mounts(rank_dir: str, volumes: List[dict],
                       volume_mounts: List[dict],
                       pvc_root: Optional[str] = None) -> List[dict]:
    """Materialize merged volumes for one rank process in the PVC-directory
    model: symlink <rank_dir>/mnt/<volume-name> -> the PVC's directory.
    Returns the mount table for KF_VOLUME_MOUNTS."""
    by_name = {v.get("name"): v for v in volumes or []}
    table = []
    for vm in volume_mounts or []:
        vol = by_name.get(vm.get("name"))
        if vol is None:
            continue
        host = None
        pvc = (vol.get("persistentVolumeClaim") or {}).get("claimName")
        if pvc and pvc_root:
            host = os.path.join(pvc_root, pvc)
        elif vol.get("hostPath"):
            host = vol["hostPath"].get("path")
        elif "emptyDir" in vol:
            host = os.path.join(rank_dir, "emptydir", vm["name"])
            os.makedirs(host, exist_ok=True)
        if host is None:
            continue
        os.makedirs(os.path.join(rank_dir, "mnt"), exist_ok=True)
        link = os.path.join(rank_dir, "mnt", vm["name"])
        if not os.path.islink(link) and not os.path.exists(link):
            os.makedirs(os.path.dirname(link), exist_ok=True)
            try:
                os.symlink(host, link)
            except FileExistsError:
                pass
        table.append({"name": vm["name"], "mountPath": vm.get("mountPath"),
                      "hostPath": host, "readOnly": bool(vm.get("readOnly")),
                      "link": link})
    return table


def mounts_env(table: List[dict]) -> Dict[str, str]:
    if not table:
        return {}
    return {"KF_VOLUME_MOUNTS": json.dumps(table)}
