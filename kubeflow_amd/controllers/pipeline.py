"""PipelineRun controller — DAG execution over gang-scheduled jobs.

The reference links out to Kubeflow Pipelines (dashboard pipeline cards,
`pipelines.kubeflow.org/enabled` namespace label —
profile_controller.go:68-73); here the DAG executor is in-repo (BASELINE
config 5: preprocess -> 4-GPU train -> 1-GPU eval -> deploy on one node).

PipelineRun.spec:
    tasks:
      - name: preprocess
        dependencies: []
        kind: PyTorchJob            # or InferenceService (deploy step)
        template: {model, steps, gpus_per_replica, replicas, ...}
    status: taskStates {name: Pending|Running|Succeeded|Failed}, conditions

Each ready task (all dependencies Succeeded) materializes as an owned
PyTorchJob / InferenceService; the gang scheduler serializes tasks whose
GPU demands overlap — gang-aware chaining on one node.
"""
from __future__ import annotations

import time
from typing import Dict, Optional

from kubeflow_amd.api import AlreadyExistsError, new_object, set_condition
from kubeflow_amd.api.objects import has_condition, owner_ref
from kubeflow_amd.controllers.base import Reconciler, RequeueAfter


class PipelineRunReconciler(Reconciler):
    kind = "PipelineRun"
    watches = ["PyTorchJob", "InferenceService"]

    def reconcile(self, namespace: Optional[str], name: str) -> None:
        run = self.store.get(self.kind, name, namespace)
        if any(has_condition(run, t) for t in ("Succeeded", "Failed")):
            return
        tasks = run["spec"].get("tasks", [])
        by_name = {t["name"]: t for t in tasks}
        states: Dict[str, str] = dict(
            run.get("status", {}).get("taskStates") or {})

        # validate DAG once
        for t in tasks:
            for dep in t.get("dependencies", []):
                if dep not in by_name:
                    set_condition(run, "Failed", "True", "InvalidDAG",
                                  f"task {t['name']} depends on unknown {dep}")
                    self.store.update(run, check_version=False)
                    return

        changed = False
        for t in tasks:
            tname = t["name"]
            state = states.get(tname, "Pending")
            if state in ("Succeeded", "Failed"):
                continue
            child_kind = t.get("kind", "PyTorchJob")
            child_name = f"{name}-{tname}"
            if state == "Pending":
                deps = t.get("dependencies", [])
                if all(states.get(d) == "Succeeded" for d in deps):
                    self._create_child(run, t, child_kind, child_name)
                    states[tname] = "Running"
                    changed = True
                continue
            # Running: check child state
            try:
                child = self.store.get(child_kind, child_name, namespace)
            except Exception:
                states[tname] = "Failed"
                changed = True
                continue
            if child_kind == "InferenceService":
                if has_condition(child, "Ready"):
                    states[tname] = "Succeeded"
                    changed = True
            elif has_condition(child, "Succeeded"):
                states[tname] = "Succeeded"
                changed = True
            elif has_condition(child, "Failed"):
                states[tname] = "Failed"
                changed = True

        run["status"]["taskStates"] = states
        if not has_condition(run, "Running"):
            set_condition(run, "Created", "True", "RunCreated", "")
            set_condition(run, "Running", "True", "RunActive", "")
            run["status"]["startTime"] = time.time()
            changed = True

        if any(s == "Failed" for s in states.values()):
            failed = [n for n, s in states.items() if s == "Failed"]
            set_condition(run, "Running", "False", "RunDone", "")
            set_condition(run, "Failed", "True", "TaskFailed",
                          f"tasks failed: {failed}")
            run["status"]["completionTime"] = time.time()
            self.store.update(run, check_version=False)
            self.store.record_event(run, "PipelineFailed", str(failed),
                                    "Warning")
            return
        if tasks and all(states.get(t["name"]) == "Succeeded" for t in tasks):
            set_condition(run, "Running", "False", "RunDone", "")
            set_condition(run, "Succeeded", "True", "AllTasksSucceeded", "")
            run["status"]["completionTime"] = time.time()
            self.store.update(run, check_version=False)
            self.store.record_event(run, "PipelineSucceeded",
                                    f"{len(tasks)} tasks")
            return
        self.store.update(run, check_version=False)
        raise RequeueAfter(0.5)

    def _create_child(self, run, task, child_kind, child_name):
        ns = run["metadata"].get("namespace")
        template = dict(task.get("template", {}))
        if child_kind == "InferenceService":
            obj = new_object(child_kind, child_name, ns,
                             spec={"predictor": template},
                             api_version="serving.kserve.io/v1beta1")
        else:
            replicas = int(template.pop("replicas", 1))
            obj = new_object(child_kind, child_name, ns, spec={
                "pytorchReplicaSpecs" if child_kind == "PyTorchJob"
                else "tfReplicaSpecs": {
                    "Worker": {"replicas": replicas,
                               "restartPolicy": template.pop(
                                   "restartPolicy", "Never"),
                               "template": template}}})
        obj["metadata"]["labels"]["pipeline-run"] = run["metadata"]["name"]
        obj["metadata"]["ownerReferences"] = [owner_ref(run)]
        try:
            self.store.create(obj)
        except AlreadyExistsError:
            pass  # re-reconcile after a crash mid-update: child exists
        self.store.record_event(run, "TaskStarted",
                                f"{task['name']} -> {child_kind}/{child_name}")
