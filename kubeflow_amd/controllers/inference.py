"""InferenceService controller — KServe-shaped CRD on the gang scheduler.

Spec (mirrors KServe's InferenceService predictor shape, the integration
point the reference enables per-namespace via the
`serving.kubeflow.org/inferenceservice` label —
profile_controller.go:68-73):

    spec:
      predictor:
        model: llama3-8b          # registry name
        gpus: 1                   # 0 = CPU (tests)
        maxBatch: 16
        maxSeqLen: 2048
        maxSlots: 16
    status:
      url: http://127.0.0.1:<port>
      conditions: [Ready, ...]

Reconcile: allocate GPU -> spawn serving_server process -> poll /healthz
until ready -> keep watching (restart on crash per restartPolicy Always,
like Deployment-backed predictors).
"""
from __future__ import annotations

import os
import urllib.request
from typing import Dict, Optional

from kubeflow_amd.api import ObjectStore, set_condition
from kubeflow_amd.api.objects import has_condition
from kubeflow_amd.controllers.base import Reconciler, RequeueAfter
from kubeflow_amd.scheduler import (GangScheduler, InsufficientResources,
                                    ProcessGang, launch_gang)
from kubeflow_amd.scheduler.launcher import free_port


class InferenceServiceReconciler(Reconciler):
    kind = "InferenceService"

    def __init__(self, store: ObjectStore, scheduler: GangScheduler,
                 serving_dir: str, volumes_dir: Optional[str] = None):
        super().__init__(store)
        self.scheduler = scheduler
        self.serving_dir = serving_dir
        self.volumes_dir = volumes_dir
        self.gangs: Dict[str, ProcessGang] = {}
        self.ports: Dict[str, int] = {}
        self.key_uid: Dict[tuple, str] = {}

    def _resolve_storage_uri(self, uri: str, namespace: Optional[str]) -> str:
        """KServe storageUri dialects in the single-node model: pvc://
        resolves against the PVC-directory root (controllers/volume.py);
        file:// and bare paths pass through; cloud schemes are rejected
        like the tensorboard logspath dialect (no object stores here).
        Raises ValueError (terminal InvalidSpec) on unusable URIs."""
        if uri.startswith("pvc://"):
            rest = uri[len("pvc://"):]
            claim, _, sub = rest.partition("/")
            if not claim or self.volumes_dir is None:
                raise ValueError(f"unresolvable storageUri {uri!r}")
            return os.path.join(self.volumes_dir, namespace or "default",
                                claim, sub)
        if uri.startswith("file://"):
            return uri[len("file://"):]
        if "://" in uri:
            raise ValueError(
                f"unsupported storageUri scheme {uri!r}: this platform "
                "serves pvc:// and local paths (no gs://, s3://)")
        return uri

    def reconcile(self, namespace: Optional[str], name: str) -> None:
        svc = self.store.get(self.kind, name, namespace)
        uid = svc["metadata"]["uid"]
        self.key_uid[(namespace, name)] = uid
        if has_condition(svc, "Failed"):  # invalid spec — terminal
            return
        gang = self.gangs.get(uid)

        if gang is None:
            self._start(svc)
            raise RequeueAfter(0.5)

        state = gang.poll()
        if state is not None:  # server process died -> restart (Always)
            self.store.record_event(svc, "PredictorCrashed",
                                    f"server exited ({state}); restarting",
                                    "Warning")
            self.gangs.pop(uid, None)
            self.scheduler.release(uid)
            set_condition(svc, "Ready", "False", "Restarting", "")
            self.store.update(svc, check_version=False)
            raise RequeueAfter(1.0)

        # readiness probe
        port = self.ports[uid]
        url = f"http://127.0.0.1:{port}"
        ready = self._probe(f"{url}/healthz")
        if ready and not has_condition(svc, "Ready"):
            set_condition(svc, "Ready", "True", "PredictorReady", url)
            svc["status"]["url"] = url
            svc["status"]["address"] = {"url": url}
            self.store.update(svc, check_version=False)
            self.store.record_event(svc, "PredictorReady", url)
        elif not ready and has_condition(svc, "Ready"):
            set_condition(svc, "Ready", "False", "ProbeFailed", "")
            self.store.update(svc, check_version=False)
        raise RequeueAfter(2.0)

    @staticmethod
    def _probe(url: str) -> bool:
        try:
            with urllib.request.urlopen(url, timeout=2) as r:
                return r.status == 200
        except Exception:
            return False

    def _start(self, svc):
        uid = svc["metadata"]["uid"]
        pred = svc["spec"].get("predictor", {})
        gpus = int(pred.get("gpus", 1))
        if gpus > 1:
            # BASELINE serving config is TP=1; multi-GPU serving (TP decode
            # + broadcast-coordinated batching) is a declared v2 seam —
            # fail loudly rather than strand an allocated-but-idle GPU.
            set_condition(svc, "Failed", "True", "InvalidSpec",
                          f"predictor.gpus={gpus}: multi-GPU serving is not "
                          "implemented in v1 (TP=1 per BASELINE config)")
            self.store.update(svc, check_version=False)
            self.store.record_event(svc, "InvalidSpec",
                                    "multi-GPU serving not implemented",
                                    "Warning")
            return
        import torch
        want_gpu = gpus > 0 and (torch.cuda.is_available()
                                 or os.environ.get("KF_FAKE_GPUS"))
        ns = svc["metadata"].get("namespace")
        if want_gpu:
            from kubeflow_amd.scheduler.quota import QuotaExceeded, admit_gpus
            try:
                admit_gpus(self.store, self.scheduler, ns, gpus)
            except QuotaExceeded as e:
                self.store.record_event(svc, "QuotaExceeded", str(e),
                                        "Warning")
                raise RequeueAfter(2.0)
        try:
            if want_gpu:
                alloc = self.scheduler.allocate(uid, gpus, namespace=ns)
                gpu_indices = alloc.gpu_indices
            else:
                self.scheduler.allocate(uid, 0)
                gpu_indices = []
        except InsufficientResources as e:
            self.store.record_event(svc, "InsufficientResources", str(e),
                                    "Warning")
            raise RequeueAfter(2.0)

        port = free_port()
        self.ports[uid] = port
        m = svc["metadata"]
        workdir = os.path.join(self.serving_dir, m.get("namespace") or
                               "default", f"{m['name']}-{uid[:8]}")
        spec = {
            "name": m["name"],
            "model": pred.get("model", "llama-tiny"),
            "port": port,
            "max_batch": pred.get("maxBatch", 16),
            "max_seq_len": pred.get("maxSeqLen", 2048),
            "max_slots": pred.get("maxSlots", 16),
            "world_size": 1,
        }
        if pred.get("storageUri"):
            try:
                spec["ckpt_dir"] = self._resolve_storage_uri(
                    pred["storageUri"], m.get("namespace"))
                spec["storage_uri"] = pred["storageUri"]
            except ValueError as e:
                self.scheduler.release(uid)
                set_condition(svc, "Failed", "True", "InvalidStorageUri",
                              str(e))
                self.store.update(svc, check_version=False)
                self.store.record_event(svc, "InvalidStorageUri", str(e),
                                        "Warning")
                return
        poddefaults = self.store.list("PodDefault", m.get("namespace"))
        try:
            gang = launch_gang(
                uid, workdir, spec, gpu_indices, poddefaults=poddefaults,
                labels=m.get("labels", {}),
                entry_module="kubeflow_amd.runtime.serving_server")
        except ValueError as e:  # spec-level launch error — terminal
            self.scheduler.release(uid)
            set_condition(svc, "Failed", "True", "InvalidSpec", str(e))
            self.store.update(svc, check_version=False)
            self.store.record_event(svc, "InvalidSpec", str(e), "Warning")
            return
        self.gangs[uid] = gang
        set_condition(svc, "Created", "True", "PredictorCreated",
                      f"port {port}")
        self.store.update(svc, check_version=False)
        self.store.record_event(svc, "SuccessfulCreate",
                                f"serving process on port {port}")

    def on_deleted(self, namespace, name):
        uid = self.key_uid.pop((namespace, name), None)
        if uid:
            gang = self.gangs.pop(uid, None)
            if gang is not None:
                gang.terminate_and_wait()
            self.scheduler.release(uid)
            self.ports.pop(uid, None)

    def shutdown(self):
        for gang in list(self.gangs.values()):
            gang.terminate_and_wait()
        self.gangs.clear()
