"""Profile controller — user namespaces as quota scopes + authz records.

Parity with profile-controller/controllers/profile_controller.go:105-312:
  * cluster-scoped Profile CR (namespace=None) owns a Namespace object with
    the workload-enabling labels (:68-73: katib-metricscollector-injection,
    serving.kubeflow.org/inferenceservice, pipelines.kubeflow.org/enabled);
  * ServiceAccounts `default-editor` / `default-viewer` (:458-537);
  * an owner RoleBinding `namespaceAdmin` binding spec.owner as admin
    (:223-244) — consumed by the kfam authz layer;
  * ResourceQuota from spec.resourceQuotaSpec (:425-455) — here it bounds
    GPU count / storage bytes for the namespace's jobs;
  * deletion cascades (ownerReferences) like the finalizer teardown.
"""
from __future__ import annotations

import os
from typing import Optional

from kubeflow_amd.api import ObjectStore, new_object, set_condition
from kubeflow_amd.api.objects import has_condition, owner_ref
from kubeflow_amd.controllers.base import Reconciler

NS_LABELS = {
    "katib-metricscollector-injection": "enabled",
    "serving.kubeflow.org/inferenceservice": "enabled",
    "pipelines.kubeflow.org/enabled": "true",
    "app.kubernetes.io/part-of": "kubeflow-profile",
}


class ProfileReconciler(Reconciler):
    kind = "Profile"

    def __init__(self, store: ObjectStore, profiles_dir: str):
        super().__init__(store)
        self.profiles_dir = profiles_dir

    def reconcile(self, namespace: Optional[str], name: str) -> None:
        prof = self.store.get(self.kind, name, None)
        owner = prof["spec"].get("owner", {})
        owner_name = (owner.get("name") if isinstance(owner, dict)
                      else str(owner)) or "anonymous@kubeflow.org"

        # namespace object (name == profile name, like the reference)
        try:
            ns = self.store.get("Namespace", name, None)
        except Exception:
            ns = new_object("Namespace", name, None, api_version="v1",
                            labels=dict(NS_LABELS))
            ns["metadata"]["ownerReferences"] = [owner_ref(prof)]
            ns["metadata"]["annotations"]["owner"] = owner_name
            self.store.create(ns)
        os.makedirs(os.path.join(self.profiles_dir, name), exist_ok=True)

        # service accounts
        for sa in ("default-editor", "default-viewer"):
            try:
                self.store.get("ServiceAccount", sa, name)
            except Exception:
                obj = new_object("ServiceAccount", sa, name, api_version="v1")
                obj["metadata"]["ownerReferences"] = [owner_ref(prof)]
                self.store.create(obj)

        # owner admin rolebinding (kfam reads these)
        try:
            self.store.get("RoleBinding", "namespaceAdmin", name)
        except Exception:
            rb = new_object(
                "RoleBinding", "namespaceAdmin", name,
                api_version="rbac.authorization.k8s.io/v1",
                annotations={"user": owner_name, "role": "admin"})
            rb["roleRef"] = {"kind": "ClusterRole", "name": "admin"}
            rb["subjects"] = [{"kind": "User", "name": owner_name}]
            rb["metadata"]["ownerReferences"] = [owner_ref(prof)]
            self.store.create(rb)

        # resource quota
        quota = prof["spec"].get("resourceQuotaSpec")
        if quota:
            try:
                q = self.store.get("ResourceQuota", "kf-resource-quota", name)
            except Exception:
                q = new_object("ResourceQuota", "kf-resource-quota", name,
                               api_version="v1", spec=quota)
                q["metadata"]["ownerReferences"] = [owner_ref(prof)]
                self.store.create(q)

        # plugins (the reference's Apply loop, profile_controller.go:262-275
        # — GCP/AWS IAM there; the local WorkloadIdentity analog + registry
        # seam here, controllers/profile_plugins.py)
        from kubeflow_amd.controllers.profile_plugins import (PluginError,
                                                              apply_plugins)
        try:
            results = apply_plugins(self.store, prof, self.profiles_dir)
            if results and prof["status"].get("plugins") != results:
                prof["status"]["plugins"] = results
                self.store.update(prof, check_version=False)
        except PluginError as e:
            set_condition(prof, "Ready", "False", "PluginFailed", str(e))
            self.store.update(prof, check_version=False)
            self.store.record_event(prof, "PluginFailed", str(e), "Warning")
            return

        if not has_condition(prof, "Ready"):
            set_condition(prof, "Ready", "True", "ProfileReady",
                          f"namespace {name} provisioned")
            self.store.update(prof, check_version=False)
            self.store.record_event(prof, "ProfileReady", name)

    def on_deleted(self, namespace, name):
        # revoke plugin state (the finalizer teardown analog,
        # profile_controller.go:277-312)
        from kubeflow_amd.controllers.profile_plugins import revoke_plugins
        try:
            revoke_plugins(self.store,
                           {"metadata": {"name": name}, "spec": {
                               "plugins": [{"kind": k}
                                           for k in ("WorkloadIdentity",)]}},
                           self.profiles_dir)
        except Exception:
            pass
