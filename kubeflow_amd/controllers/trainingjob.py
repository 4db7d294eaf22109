"""PyTorchJob / TFJob controller — gang-scheduled distributed training.

Replaces the sibling-repo training-operator the reference integrates with
(SURVEY.md §2.12): same CRD shape —

    spec.pytorchReplicaSpecs.{Master,Worker}.replicas     (tfReplicaSpecs for
    spec....restartPolicy: Never|OnFailure                 TFJob)
    spec....template: the worker spec {model, steps, micro_batch, seq_len,
                      lr, gpus_per_replica, ...}
    status.conditions[]: Created/Running/Succeeded/Failed/Restarting
    status.replicaStatuses.{Master,Worker}: {active, succeeded, failed}

— but reconciled onto a gang of one-process-per-GPU workers over RCCL/xGMI
instead of pods. Worker heartbeats (rank-*/status.json) feed job status the
way pod state + events feed notebook status in the reference
(apps/common/status.py:10-99).
"""
from __future__ import annotations

import json
import os
import time
from typing import Dict, Optional

from kubeflow_amd.api import ObjectStore, set_condition
from kubeflow_amd.api.objects import has_condition
from kubeflow_amd.controllers.base import Reconciler, RequeueAfter
from kubeflow_amd.scheduler import (GangScheduler, InsufficientResources,
                                    ProcessGang, launch_gang)

TERMINAL = ("Succeeded", "Failed")


class TrainingJobReconciler(Reconciler):
    kind = "PyTorchJob"
    replica_field = "pytorchReplicaSpecs"

    def __init__(self, store: ObjectStore, scheduler: GangScheduler,
                 jobs_dir: str, poll_period: float = 0.25, warm_pool=None):
        super().__init__(store)
        self.scheduler = scheduler
        self.jobs_dir = jobs_dir
        self.poll_period = poll_period
        self.warm_pool = warm_pool
        self.gangs: Dict[str, ProcessGang] = {}
        self.eval_gangs: Dict[str, ProcessGang] = {}
        self.restarts: Dict[str, int] = {}
        self.key_uid: Dict[tuple, str] = {}

    # ------------------------------------------------------------- helpers
    # replica-type semantics (training-operator parity): Master/Chief is
    # rank 0 and its template wins; Workers follow; Evaluator runs as a
    # checkpoint-watching sidecar OUTSIDE the gang; PS has no analog in a
    # torch runtime and is rejected loudly.
    ROLE_ORDER = ("Master", "Chief", "Worker")

    def _replicas(self, job) -> int:
        specs = job["spec"].get(self.replica_field, {})
        return sum(int(r.get("replicas", 1)) for role, r in specs.items()
                   if role != "Evaluator")

    def _role_of_rank(self, job, rank: int) -> str:
        specs = job["spec"].get(self.replica_field, {})
        off = 0
        for role in self.ROLE_ORDER:
            if role in specs:
                n = int(specs[role].get("replicas", 1))
                if rank < off + n:
                    return role
                off += n
        return "Worker"

    def _evaluator_spec(self, job):
        specs = job["spec"].get(self.replica_field, {})
        return specs.get("Evaluator")

    def _template(self, job) -> dict:
        specs = job["spec"].get(self.replica_field, {})
        for role in ("Master", "Chief", "Worker"):
            if role in specs and "template" in specs[role]:
                return dict(specs[role]["template"])
        return {}

    def _restart_policy(self, job) -> str:
        specs = job["spec"].get(self.replica_field, {})
        for r in specs.values():
            if "restartPolicy" in r:
                return r["restartPolicy"]
        return "Never"

    def _workdir(self, job) -> str:
        m = job["metadata"]
        return os.path.join(self.jobs_dir, m.get("namespace") or "default",
                            f"{m['name']}-{m['uid'][:8]}")

    # ----------------------------------------------------------- reconcile
    def reconcile(self, namespace: Optional[str], name: str) -> None:
        job = self.store.get(self.kind, name, namespace)
        uid = job["metadata"]["uid"]
        self.key_uid[(namespace, name)] = uid

        if any(has_condition(job, t) for t in TERMINAL):
            self._cleanup(uid)
            return

        gang = self.gangs.get(uid)
        if gang is None:
            self._start(job)
            raise RequeueAfter(self.poll_period)

        self._sync_status(job, gang)
        raise RequeueAfter(self.poll_period)

    def _start(self, job):
        uid = job["metadata"]["uid"]
        specs = job["spec"].get(self.replica_field, {})
        if "PS" in specs:
            set_condition(job, "Failed", "True", "InvalidSpec",
                          "PS replicas: parameter-server strategy has no "
                          "analog in the torch runtime (use DDP Workers; "
                          "tf.distribute PS-strategy is not supported)")
            self.store.update(job, check_version=False)
            self.store.record_event(job, "InvalidSpec",
                                    "PS replicas unsupported", "Warning")
            return
        n = self._replicas(job)
        template = self._template(job)
        gpus_per = int(template.get("gpus_per_replica", 1))
        import torch
        want_gpu = gpus_per > 0 and (
            torch.cuda.is_available() or os.environ.get("KF_FAKE_GPUS"))
        ns = job["metadata"].get("namespace")
        if want_gpu:
            from kubeflow_amd.scheduler.quota import QuotaExceeded, admit_gpus
            try:
                admit_gpus(self.store, self.scheduler, ns, n * gpus_per)
            except QuotaExceeded as e:
                # quota rejections requeue (a peer releasing frees budget),
                # surfaced as Warning events like FailedCreate
                self.store.record_event(job, "QuotaExceeded", str(e),
                                        "Warning")
                if not has_condition(job, "Created"):
                    set_condition(job, "Created", "True", "JobCreated",
                                  "blocked by ResourceQuota")
                    self.store.update(job, check_version=False)
                raise RequeueAfter(2.0)
        # shared GPU allocation (HBM-accounted co-scheduling) for small
        # workloads like HPO trials: template {gpu_shared: true,
        # gpu_memory: "24Gi"} packs multiple jobs per GPU
        shared = bool(template.get("gpu_shared"))
        mem = 0
        if shared:
            v = str(template.get("gpu_memory", "24Gi"))
            units = {"Ki": 1 << 10, "Mi": 1 << 20, "Gi": 1 << 30}
            mem = next((int(float(v[:-2]) * m) for sfx, m in units.items()
                        if v.endswith(sfx)), 24 << 30)
        try:
            if want_gpu:
                alloc = self.scheduler.allocate(uid, n * gpus_per,
                                                exclusive=not shared,
                                                mem_per_gpu=mem,
                                                namespace=ns)
                gpu_indices = alloc.gpu_indices
            else:
                self.scheduler.allocate(uid, 0)
                gpu_indices = []
        except InsufficientResources as e:
            if not has_condition(job, "Created"):
                set_condition(job, "Created", "True", "JobCreated",
                              "waiting for GPUs")
                self.store.update(job, check_version=False)
            self.store.record_event(job, "InsufficientResources", str(e),
                                    "Warning")
            raise RequeueAfter(2.0)

        workdir = self._workdir(job)
        spec = dict(template)
        spec.setdefault("model", "mnist-mlp")
        spec["world_size"] = n
        poddefaults = self.store.list("PodDefault",
                                      job["metadata"].get("namespace"))
        configmaps = {c["metadata"]["name"]: c.get("data", {})
                      for c in self.store.list(
                          "ConfigMap", job["metadata"].get("namespace"))}
        numa = {g.index: g.numa_node for g in self.scheduler.inv.gpus}
        try:
            gang = launch_gang(uid, workdir, spec, gpu_indices,
                               poddefaults=poddefaults,
                               labels=job["metadata"].get("labels", {}),
                               numa_nodes=numa, warm_pool=self.warm_pool,
                               configmaps=configmaps)
        except ValueError as e:
            # spec-level launch error (e.g. PodDefault env conflict) —
            # terminal, not retryable: mark Failed instead of hot-looping
            self.scheduler.release(uid)
            set_condition(job, "Failed", "True", "InvalidSpec", str(e))
            self.store.update(job, check_version=False)
            self.store.record_event(job, "InvalidSpec", str(e), "Warning")
            return
        self.gangs[uid] = gang
        ev = self._evaluator_spec(job)
        if ev is not None:
            ev_spec = dict(template)
            ev_spec.update(ev.get("template", {}))
            ev_spec["role"] = "Evaluator"
            ev_spec["world_size"] = 1
            ev_spec.setdefault("checkpoint_dir",
                               template.get("checkpoint_dir")
                               or os.path.join(workdir, "checkpoints"))
            try:
                self.eval_gangs[uid] = launch_gang(
                    uid + "-eval", os.path.join(workdir, "evaluator"),
                    ev_spec, [], poddefaults=poddefaults,
                    labels=job["metadata"].get("labels", {}))
            except ValueError as e:
                self.store.record_event(job, "EvaluatorFailed", str(e),
                                        "Warning")
        set_condition(job, "Created", "True", "JobCreated", "gang launched")
        set_condition(job, "Running", "True", "JobRunning",
                      f"{n} replicas on GPUs {gpu_indices or 'cpu'}")
        # an OnFailure relaunch is no longer restarting once Running flips
        # back on (training-operator flips Restarting=False on the
        # Restarting->Running transition)
        if has_condition(job, "Restarting"):
            set_condition(job, "Restarting", "False", "JobRunning",
                          "restart complete")
        job["status"]["startTime"] = job["status"].get("startTime") or time.time()
        self.store.update(job, check_version=False)
        self.store.record_event(job, "SuccessfulCreate",
                                f"created gang of {n} workers")

    def _read_rank_status(self, gang: ProcessGang) -> Dict[int, dict]:
        out = {}
        for r in gang.ranks:
            path = os.path.join(gang.workdir, f"rank-{r.rank}", "status.json")
            try:
                with open(path) as f:
                    out[r.rank] = json.load(f)
            except (OSError, json.JSONDecodeError):
                out[r.rank] = {"state": "initializing"}
        return out

    def _sync_status(self, job, gang: ProcessGang):
        uid = job["metadata"]["uid"]
        state = gang.poll()
        ranks = self._read_rank_status(gang)
        by_role: Dict[str, Dict[str, int]] = {}
        for rk, st in ranks.items():
            role = self._role_of_rank(job, rk)
            slot = by_role.setdefault(role, {"active": 0, "succeeded": 0,
                                             "failed": 0})
            key = {"running": "active", "succeeded": "succeeded",
                   "failed": "failed"}.get(st.get("state"), "active")
            slot[key] += 1
        ev_gang = self.eval_gangs.get(uid)
        if ev_gang is not None:
            ev_state = ev_gang.poll()
            ev_rank = self._read_rank_status(ev_gang).get(0, {})
            job["status"]["replicaStatuses"] = job["status"].get(
                "replicaStatuses", {})
            by_role["Evaluator"] = {
                "active": 1 if ev_state is None else 0,
                "succeeded": 1 if ev_state == "Succeeded" else 0,
                "failed": 1 if ev_state == "Failed" else 0}
            if ev_rank.get("metrics", {}).get("eval_loss") is not None:
                job["status"]["evalMetrics"] = ev_rank["metrics"]
        job["status"]["replicaStatuses"] = by_role
        r0 = ranks.get(0, {})
        if r0.get("metrics"):
            job["status"]["trainingMetrics"] = r0["metrics"]
            job["status"]["step"] = r0.get("step", 0)

        if state == "Succeeded":
            set_condition(job, "Running", "False", "JobFinished", "")
            set_condition(job, "Succeeded", "True", "JobSucceeded",
                          "all ranks exited 0")
            job["status"]["completionTime"] = time.time()
            self.store.record_event(job, "JobSucceeded", "training complete")
            self.scheduler.release(uid)
            self.gangs.pop(uid, None)
        elif state == "Failed":
            policy = self._restart_policy(job)
            limit = int(job["spec"].get("backoffLimit", 3))
            nrestart = self.restarts.get(uid, 0)
            err = next((s.get("error") for s in ranks.values()
                        if s.get("error")), "worker exited nonzero")
            if policy == "OnFailure" and nrestart < limit:
                self.restarts[uid] = nrestart + 1
                self.scheduler.release(uid)
                self.gangs.pop(uid, None)
                set_condition(job, "Restarting", "True", "JobRestarting",
                              f"restart {nrestart + 1}/{limit}: {err}")
                self.store.record_event(job, "JobRestarting", err, "Warning")
            else:
                set_condition(job, "Running", "False", "JobFinished", "")
                set_condition(job, "Failed", "True", "JobFailed", err)
                job["status"]["completionTime"] = time.time()
                self.store.record_event(job, "JobFailed", err, "Warning")
                self.scheduler.release(uid)
                self.gangs.pop(uid, None)
        self.store.update(job, check_version=False)

    def _cleanup(self, uid: str):
        gang = self.gangs.pop(uid, None)
        if gang is not None:
            gang.terminate_and_wait()
        ev = self.eval_gangs.pop(uid, None)
        if ev is not None:
            ev.terminate_and_wait()
        self.scheduler.release(uid)

    def on_deleted(self, namespace, name):
        uid = self.key_uid.pop((namespace, name), None)
        if uid:
            self._cleanup(uid)

    def shutdown(self):
        for gang in list(self.gangs.values()):
            gang.terminate_and_wait()
        self.gangs.clear()


class TFJobReconciler(TrainingJobReconciler):
    """TFJob parity: same reconcile machinery, tfReplicaSpecs field.
    (The reference treats TFJob/PyTorchJob as parallel integrations —
    testing/kf_is_ready_test.py:99-113 expects both operators.)"""
    kind = "TFJob"
    replica_field = "tfReplicaSpecs"
