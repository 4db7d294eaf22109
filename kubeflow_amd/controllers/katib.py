"""Katib Experiment/Trial controllers — HPO with GPU bin-packing.

CRD shapes follow Katib v1beta1 (the reference smoke-tests StudyJobs — the
v1alpha1 ancestor — via the custom-objects API,
testing/katib_studyjob_test.py:39-120):

Experiment.spec:
    objective: {type: minimize|maximize, objectiveMetricName: loss, goal: f}
    algorithm: {algorithmName: random|grid|bayesianoptimization}
    parallelTrialCount / maxTrialCount / maxFailedTrialCount
    parameters: [{name, parameterType, feasibleSpace}, ...]
    trialTemplate: worker spec (model/steps/gpus_per_replica/...) whose
        fields named by `parameters` are overridden per-trial
Experiment.status: conditions, trialsRunning/Succeeded/Failed,
    currentOptimalTrial {parameterAssignments, observation}

Each Trial owns one PyTorchJob (1 replica, gpus_per_replica GPUs) — the gang
scheduler bin-packs one trial per GPU, so parallelTrialCount=8 fills the
node (BASELINE config 4).
"""
from __future__ import annotations

import time
from typing import Dict, List, Optional, Tuple

from kubeflow_amd.api import ObjectStore, new_object, set_condition
from kubeflow_amd.api.objects import has_condition, owner_ref, get_condition
from kubeflow_amd.controllers.base import Reconciler, RequeueAfter
from kubeflow_amd.katib import make_suggestion


class ExperimentReconciler(Reconciler):
    kind = "Experiment"
    watches = ["Trial"]

    def __init__(self, store: ObjectStore):
        super().__init__(store)
        self._suggesters: Dict[str, object] = {}
        self._key_uid: Dict[tuple, str] = {}

    def on_deleted(self, namespace, name):
        uid = self._key_uid.pop((namespace, name), None)
        if uid:
            self._suggesters.pop(uid, None)

    def reconcile(self, namespace: Optional[str], name: str) -> None:
        exp = self.store.get(self.kind, name, namespace)
        self._key_uid[(namespace, name)] = exp["metadata"]["uid"]
        if any(has_condition(exp, t) for t in ("Succeeded", "Failed")):
            return
        spec = exp["spec"]
        uid = exp["metadata"]["uid"]
        parallel = int(spec.get("parallelTrialCount", 1))
        max_trials = int(spec.get("maxTrialCount", 8))
        max_failed = int(spec.get("maxFailedTrialCount", 3))
        objective = spec.get("objective", {})
        metric = objective.get("objectiveMetricName", "loss")
        minimize = objective.get("type", "minimize") == "minimize"
        goal = objective.get("goal")

        trials = self.store.list("Trial", namespace,
                                 {"experiment": name})
        running = [t for t in trials
                   if not any(has_condition(t, c)
                              for c in ("Succeeded", "Failed"))]
        succeeded = [t for t in trials if has_condition(t, "Succeeded")]
        failed = [t for t in trials if has_condition(t, "Failed")]

        # observations for the suggester + optimal trial
        observed: List[Tuple[dict, Optional[float]]] = []
        best = None
        for t in trials:
            obs = t.get("status", {}).get("observation")
            val = None
            if obs is not None:
                val = obs.get("value")
            assignment = {a["name"]: a["value"]
                          for a in t["spec"].get("parameterAssignments", [])}
            sval = val if (val is None or minimize) else -val
            observed.append((assignment, sval))
            if val is not None and (
                    best is None or
                    (minimize and val < best[1]) or
                    (not minimize and val > best[1])):
                best = (t, val)

        exp["status"].update({
            "trials": len(trials),
            "trialsRunning": len(running),
            "trialsSucceeded": len(succeeded),
            "trialsFailed": len(failed),
        })
        if best is not None:
            t, val = best
            exp["status"]["currentOptimalTrial"] = {
                "bestTrialName": t["metadata"]["name"],
                "parameterAssignments": t["spec"].get("parameterAssignments"),
                "observation": {"metrics": [{"name": metric, "latest": val}]},
            }

        goal_met = (best is not None and goal is not None and
                    ((minimize and best[1] <= goal) or
                     (not minimize and best[1] >= goal)))

        # upstream semantics: experiment fails when the failure count
        # REACHES maxFailedTrialCount (>=, not >)
        if len(failed) >= max_failed and max_failed > 0:
            set_condition(exp, "Failed", "True", "TooManyFailedTrials",
                          f"{len(failed)} trials failed")
            self.store.update(exp, check_version=False)
            self.store.record_event(exp, "ExperimentFailed",
                                    "maxFailedTrialCount exceeded", "Warning")
            return

        if (goal_met or len(trials) >= max_trials) and not running:
            set_condition(exp, "Running", "False", "ExperimentDone", "")
            set_condition(exp, "Succeeded", "True",
                          "GoalReached" if goal_met else "MaxTrialsReached",
                          f"best {metric}={None if best is None else best[1]}")
            self.store.update(exp, check_version=False)
            self.store.record_event(exp, "ExperimentSucceeded",
                                    str(exp["status"].get("currentOptimalTrial")))
            return

        # propose new trials up to parallelism / budget
        budget = min(parallel - len(running), max_trials - len(trials))
        if budget > 0 and not goal_met:
            sug = self._suggesters.get(uid)
            if sug is None:
                sug = make_suggestion(
                    spec.get("algorithm", {}).get("algorithmName", "random"),
                    spec.get("parameters", []), seed=hashd(uid))
                self._suggesters[uid] = sug
            for assignment in sug.suggest(observed, budget):
                self._create_trial(exp, assignment, len(trials))
                trials.append(None)  # count only

        if not has_condition(exp, "Running"):
            set_condition(exp, "Created", "True", "ExperimentCreated", "")
            set_condition(exp, "Running", "True", "ExperimentRunning", "")
        self.store.update(exp, check_version=False)
        raise RequeueAfter(1.0)

    def _create_trial(self, exp, assignment: dict, index: int):
        name = f"{exp['metadata']['name']}-trial-{index}-{int(time.time()*1000)%100000}"
        trial = new_object(
            "Trial", name, exp["metadata"].get("namespace"),
            labels={"experiment": exp["metadata"]["name"]},
            spec={
                "parameterAssignments": [
                    {"name": k, "value": v} for k, v in assignment.items()],
                "template": exp["spec"].get("trialTemplate", {}),
                "objectiveMetricName": exp["spec"].get("objective", {}).get(
                    "objectiveMetricName", "loss"),
            })
        trial["metadata"]["ownerReferences"] = [owner_ref(exp)]
        self.store.create(trial)
        self.store.record_event(exp, "TrialCreated", name)


def hashd(s: str) -> int:
    import hashlib
    return int(hashlib.sha1(s.encode()).hexdigest()[:8], 16)


class TrialReconciler(Reconciler):
    kind = "Trial"
    watches = ["PyTorchJob"]

    def reconcile(self, namespace: Optional[str], name: str) -> None:
        trial = self.store.get(self.kind, name, namespace)
        if any(has_condition(trial, t) for t in ("Succeeded", "Failed")):
            return
        jobname = f"{name}-job"
        try:
            job = self.store.get("PyTorchJob", jobname, namespace)
        except Exception:
            job = self._create_job(trial, jobname)
            set_condition(trial, "Running", "True", "TrialRunning", jobname)
            self.store.update(trial, check_version=False)
            raise RequeueAfter(1.0)

        metric_name = trial["spec"].get("objectiveMetricName", "loss")
        metrics = job.get("status", {}).get("trainingMetrics", {})
        if metrics.get(metric_name) is not None:
            trial["status"]["observation"] = {
                "metric": metric_name, "value": metrics[metric_name]}
        if has_condition(job, "Succeeded"):
            set_condition(trial, "Running", "False", "TrialDone", "")
            set_condition(trial, "Succeeded", "True", "TrialSucceeded",
                          f"{metric_name}={metrics.get(metric_name)}")
            self.store.update(trial, check_version=False)
            return
        if has_condition(job, "Failed"):
            set_condition(trial, "Running", "False", "TrialDone", "")
            set_condition(trial, "Failed", "True", "TrialFailed",
                          get_condition(job, "Failed").get("message", ""))
            self.store.update(trial, check_version=False)
            return
        self.store.update(trial, check_version=False)
        raise RequeueAfter(1.0)

    def _create_job(self, trial, jobname):
        template = dict(trial["spec"].get("template", {}))
        for a in trial["spec"].get("parameterAssignments", []):
            template[a["name"]] = a["value"]
        job = new_object(
            "PyTorchJob", jobname, trial["metadata"].get("namespace"),
            labels={"trial": trial["metadata"]["name"]},
            spec={"pytorchReplicaSpecs": {"Worker": {
                "replicas": int(template.pop("replicas", 1)),
                "restartPolicy": "Never",
                "template": template,
            }}})
        job["metadata"]["ownerReferences"] = [owner_ref(trial)]
        return self.store.create(job)
