"""Profile plugin seam — the reference's Plugin interface
(profile_controller.go:78-84: `ApplyPlugin(ctx, profile) / RevokePlugin`)
with its two cloud implementations mapped to the single-node model:

  * reference `KIND_WORKLOAD_IDENTITY` (plugin_workload_identity.go:32-44)
    annotates the default-editor ServiceAccount with a GCP service
    account and edits cloud IAM policy;
  * reference `KIND_AWS_IAM_FOR_SERVICE_ACCOUNT` (plugin_iam.go:22-33)
    annotates the SA with a role ARN and updates the trust policy.

There is no cloud IAM on one MI355X node, so the shipped plugin is the
local analog of both: `WorkloadIdentity` binds the namespace's
default-editor ServiceAccount to a LOCAL identity — it annotates the SA
(the same seam the cloud plugins use) and materializes a credential file
under the profile's directory that workloads can mount. The registry is
the extension point: a real cloud plugin drops in as another entry with
apply/revoke, exactly like the reference's switch on plugin kind
(profile_controller.go:541-575 GetPluginSpec + Apply loop).

Profile spec shape (reference-compatible):
    spec:
      plugins:
        - kind: WorkloadIdentity
          spec: {identity: "svc-team-a@local"}
"""
from __future__ import annotations

import json
import os
from typing import Callable, Dict


class PluginError(Exception):
    pass


def _apply_workload_identity(store, profile, spec, profiles_dir: str):
    ns = profile["metadata"]["name"]
    identity = spec.get("identity") or f"{ns}@local"
    sa = store.get("ServiceAccount", "default-editor", ns)
    ann = sa["metadata"].setdefault("annotations", {})
    # the same annotation seam the GCP plugin uses (iam.gke.io/gcp-service-
    # account); local flavor carries a local identity name
    if ann.get("iam.kubeflow.org/local-identity") != identity:
        ann["iam.kubeflow.org/local-identity"] = identity
        store.update(sa, check_version=False)
    cred_dir = os.path.join(profiles_dir, ns)
    os.makedirs(cred_dir, exist_ok=True)
    cred = os.path.join(cred_dir, "identity.json")
    payload = {"identity": identity, "namespace": ns,
               "serviceAccount": "default-editor"}
    if not os.path.exists(cred) or json.load(open(cred)) != payload:
        with open(cred, "w") as f:
            json.dump(payload, f)
    return {"identity": identity, "credentialPath": cred}


def _revoke_workload_identity(store, profile, spec, profiles_dir: str):
    ns = profile["metadata"]["name"]
    try:
        sa = store.get("ServiceAccount", "default-editor", ns)
        if sa["metadata"].get("annotations", {}).pop(
                "iam.kubeflow.org/local-identity", None) is not None:
            store.update(sa, check_version=False)
    except Exception:
        pass
    cred = os.path.join(profiles_dir, ns, "identity.json")
    if os.path.exists(cred):
        os.remove(cred)


# kind -> (apply, revoke); a cloud plugin registers here
PLUGINS: Dict[str, Dict[str, Callable]] = {
    "WorkloadIdentity": {"apply": _apply_workload_identity,
                         "revoke": _revoke_workload_identity},
}


def apply_plugins(store, profile, profiles_dir: str) -> Dict[str, dict]:
    """Apply every spec.plugins entry (the reference's Apply loop,
    profile_controller.go:262-275). Unknown kinds raise PluginError —
    matching the reference's error on an unhandled plugin spec."""
    results = {}
    for p in profile["spec"].get("plugins") or []:
        kind = p.get("kind")
        impl = PLUGINS.get(kind)
        if impl is None:
            raise PluginError(f"unknown profile plugin kind {kind!r} "
                              f"(registered: {sorted(PLUGINS)})")
        results[kind] = impl["apply"](store, profile, p.get("spec") or {},
                                      profiles_dir)
    return results


def revoke_plugins(store, profile, profiles_dir: str) -> None:
    for p in profile["spec"].get("plugins") or []:
        impl = PLUGINS.get(p.get("kind"))
        if impl is not None:
            impl["revoke"](store, profile, p.get("spec") or {},
                           profiles_dir)
