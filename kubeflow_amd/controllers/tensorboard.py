"""Tensorboard controller — metrics-viewer sessions per Tensorboard CR.

Parity with the reference tensorboard-controller
(controllers/tensorboard_controller.go):
  * spec.logspath dialects (path parsing :344-374): `pvc://<name>/<sub>`
    resolves against the platform's volumes directory (the PVC analog),
    `file://<abs>` and plain paths are used as-is; `gs://`/`s3://` are
    rejected (no cloud in a single-node deployment);
  * Deployment+Service+VirtualService collapse into one process and a
    status.url;
  * RWO co-scheduling concerns (generateNodeAffinity :392) vanish — all
    storage is node-local by construction.
"""
from __future__ import annotations

import json
import os
import subprocess
import sys
import urllib.request
from typing import Dict, Optional

from kubeflow_amd.api import ObjectStore, set_condition
from kubeflow_amd.api.objects import has_condition
from kubeflow_amd.controllers.base import Reconciler, RequeueAfter
from kubeflow_amd.scheduler.launcher import free_port, _preexec


class TensorboardReconciler(Reconciler):
    kind = "Tensorboard"

    def __init__(self, store: ObjectStore, tb_dir: str,
                 volumes_dir: Optional[str] = None):
        super().__init__(store)
        self.tb_dir = tb_dir
        self.volumes_dir = volumes_dir or os.path.join(
            os.path.dirname(tb_dir), "volumes")
        self.sessions: Dict[str, tuple] = {}
        self.key_uid: Dict[tuple, str] = {}

    def resolve_logspath(self, namespace: str, logspath: str) -> str:
        if logspath.startswith("pvc://"):
            rest = logspath[len("pvc://"):]
            pvc, _, sub = rest.partition("/")
            return os.path.join(self.volumes_dir, namespace or "default",
                                pvc, sub)
        if logspath.startswith("file://"):
            return logspath[len("file://"):]
        if logspath.startswith(("gs://", "s3://")):
            raise ValueError(
                f"cloud logspath {logspath!r} unsupported on single-node")
        return logspath

    def reconcile(self, namespace: Optional[str], name: str) -> None:
        tb = self.store.get(self.kind, name, namespace)
        uid = tb["metadata"]["uid"]
        self.key_uid[(namespace, name)] = uid
        sess = self.sessions.get(uid)
        if sess is None or sess[0].poll() is not None:
            try:
                logdir = self.resolve_logspath(
                    namespace, tb["spec"].get("logspath", ""))
            except ValueError as e:
                set_condition(tb, "Failed", "True", "BadLogsPath", str(e))
                self.store.update(tb, check_version=False)
                return
            self._start(tb, logdir)
            raise RequeueAfter(0.5)
        port = sess[1]
        url = f"http://127.0.0.1:{port}"
        ready = self._probe(url + "/healthz")
        tb["status"]["readyReplicas"] = 1 if ready else 0
        tb["status"]["url"] = f"{url}/tensorboard/{namespace}/{name}/"
        if ready and not has_condition(tb, "Running"):
            set_condition(tb, "Running", "True", "Ready", url)
            self.store.record_event(tb, "Started", url)
        self.store.update(tb, check_version=False)
        raise RequeueAfter(0.3 if not ready else 5.0)

    @staticmethod
    def _probe(url):
        try:
            with urllib.request.urlopen(url, timeout=2) as r:
                return r.status == 200
        except Exception:
            return False

    def _start(self, tb, logdir):
        uid = tb["metadata"]["uid"]
        m = tb["metadata"]
        ns = m.get("namespace") or "default"
        port = free_port()
        workdir = os.path.join(self.tb_dir, ns, m["name"])
        os.makedirs(workdir, exist_ok=True)
        os.makedirs(logdir, exist_ok=True)
        spec_path = os.path.join(workdir, "tb.json")
        with open(spec_path, "w") as f:
            json.dump({"port": port, "logdir": logdir}, f)
        env = dict(os.environ)
        repo_root = os.path.dirname(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))))
        env["PYTHONPATH"] = repo_root + os.pathsep + env.get("PYTHONPATH", "")
        logf = open(os.path.join(workdir, "tb.log"), "w")
        proc = subprocess.Popen(
            [sys.executable, "-m", "kubeflow_amd.runtime.tensorboard_server",
             "--spec", spec_path],
            env=env, stdout=logf, stderr=logf, cwd=workdir,
            preexec_fn=_preexec)
        self.sessions[uid] = (proc, port)
        set_condition(tb, "Created", "True", "SessionCreated", f"port {port}")
        self.store.update(tb, check_version=False)

    def on_deleted(self, namespace, name):
        uid = self.key_uid.pop((namespace, name), None)
        sess = self.sessions.pop(uid, None) if uid else None
        if sess and sess[0].poll() is None:
            sess[0].terminate()

    def shutdown(self):
        for proc, _ in self.sessions.values():
            if proc.poll() is None:
                proc.terminate()
        self.sessions.clear()
