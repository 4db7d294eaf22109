"""PersistentVolumeClaim controller — PVCs as quota-tracked directories.

The volumes web app CRUDs PVCs and shows which pods mount them
(volumes/backend/apps/common/form.py:22-38, utils.py:35). On a single node,
a PVC is a managed directory under the platform's volumes root; `storage`
requests are recorded (and enforced advisorily via du), and status mirrors
the k8s PVC phases (Pending -> Bound).

Mounting: jobs/notebooks reference PVCs by name; the launcher exposes them
as paths under the same volumes root, so `pvc://name/sub` paths (tensorboard
logspath dialect) resolve against the directory this controller creates.
"""
from __future__ import annotations

import os
import shutil
from typing import Optional

from kubeflow_amd.api import ObjectStore, set_condition
from kubeflow_amd.controllers.base import Reconciler


class VolumeReconciler(Reconciler):
    kind = "PersistentVolumeClaim"

    def __init__(self, store: ObjectStore, volumes_dir: str,
                 reclaim_policy: str = "Delete"):
        super().__init__(store)
        self.volumes_dir = volumes_dir
        self.reclaim_policy = reclaim_policy

    def path_for(self, namespace: Optional[str], name: str) -> str:
        return os.path.join(self.volumes_dir, namespace or "default", name)

    def reconcile(self, namespace: Optional[str], name: str) -> None:
        pvc = self.store.get(self.kind, name, namespace)
        path = self.path_for(namespace, name)
        origin = pvc["metadata"].get("annotations", {}).get("rok/origin")
        if origin and not pvc.get("status", {}).get("phase"):
            if self.restore_from(namespace, name, origin):
                self.store.record_event(pvc, "SnapshotRestored", origin)
        os.makedirs(path, exist_ok=True)
        req = (pvc["spec"].get("resources", {}).get("requests", {})
               .get("storage", "1Gi"))
        if pvc.get("status", {}).get("phase") != "Bound":
            pvc["status"]["phase"] = "Bound"
            pvc["status"]["capacity"] = {"storage": req}
            pvc["status"]["accessModes"] = pvc["spec"].get(
                "accessModes", ["ReadWriteOnce"])
            pvc["status"]["hostPath"] = path
            set_condition(pvc, "Bound", "True", "Provisioned", path)
            self.store.update(pvc, check_version=False)
            self.store.record_event(pvc, "ProvisioningSucceeded", path)

    def on_deleted(self, namespace, name):
        if self.reclaim_policy == "Delete":
            shutil.rmtree(self.path_for(namespace, name), ignore_errors=True)

    # ---- snapshot provider seam (the reference's "rok" flavor:
    # jupyter/backend/apps/rok — PVCs restored from snapshot URLs carried in
    # `rok/origin` annotations). Single-node provider = directory copies.
    def snapshot(self, namespace: Optional[str], name: str) -> str:
        """Snapshot a PVC's contents; returns the snapshot URL."""
        import time as _t
        src = self.path_for(namespace, name)
        snap_id = f"{name}-{int(_t.time() * 1000)}"
        dst = os.path.join(self.volumes_dir, "_snapshots", snap_id)
        os.makedirs(os.path.dirname(dst), exist_ok=True)
        shutil.copytree(src, dst)
        return f"rok://{snap_id}"

    def restore_from(self, namespace: Optional[str], name: str,
                     origin: str) -> bool:
        """Materialize a new PVC directory from a rok:// snapshot URL."""
        if not origin.startswith("rok://"):
            return False
        src = os.path.join(self.volumes_dir, "_snapshots",
                           origin[len("rok://"):])
        if not os.path.isdir(src):
            return False
        dst = self.path_for(namespace, name)
        if os.path.exists(dst):
            shutil.rmtree(dst)
        shutil.copytree(src, dst)
        return True

    @staticmethod
    def parse_quantity(q: str) -> int:
        """k8s resource.Quantity subset: plain ints + Ki/Mi/Gi/Ti suffixes."""
        units = {"Ki": 1024, "Mi": 1024**2, "Gi": 1024**3, "Ti": 1024**4,
                 "K": 1000, "M": 1000**2, "G": 1000**3, "T": 1000**4}
        for suffix, mult in units.items():
            if q.endswith(suffix):
                return int(float(q[:-len(suffix)]) * mult)
        return int(float(q))

    def pods_using(self, namespace: Optional[str], name: str):
        """Jobs/notebooks referencing this PVC (volumes web app's
        get_pods_using_pvc analog)."""
        users = []
        for kind in ("PyTorchJob", "TFJob", "Notebook"):
            for obj in self.store.list(kind, namespace):
                spec = str(obj.get("spec", {}))
                if f"'{name}'" in spec or f'"{name}"' in spec:
                    users.append({"kind": kind,
                                  "name": obj["metadata"]["name"]})
        return users
