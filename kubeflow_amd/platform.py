"""Platform assembly — the single-node "cluster".

Wires the object store, the gang scheduler and every controller into one
process, the way the reference wires its controller managers + webapps onto
a Kubernetes cluster (SURVEY.md §1 layer map, collapsed).

    plat = Platform(root_dir="/var/lib/kubeflow-amd")
    plat.start()
    plat.store.create(new_object("PyTorchJob", "train-1", spec={...}))
    ...
    plat.stop()
"""
from __future__ import annotations

import os
import tempfile
from typing import Optional

from kubeflow_amd.api import ObjectStore
from kubeflow_amd.controllers.base import ControllerManager
from kubeflow_amd.scheduler import GangScheduler, GpuInventory


class Platform:
    def __init__(self, root_dir: Optional[str] = None, persist: bool = False):
        self.root_dir = root_dir or tempfile.mkdtemp(prefix="kubeflow-amd-")
        os.makedirs(self.root_dir, exist_ok=True)
        self.store = ObjectStore(
            persist_path=os.path.join(self.root_dir, "store.jsonl")
            if persist else None)
        self.inventory = GpuInventory()
        self.scheduler = GangScheduler(self.inventory)
        from kubeflow_amd.scheduler.warmpool import WarmPool
        pool_size = int(os.environ.get(
            "KF_WARM_POOL", str(max(2, self.inventory.n_gpus))))
        self.warm_pool = (WarmPool(pool_size,
                                   os.path.join(self.root_dir, "warmpool"))
                          if pool_size > 0 else None)
        self.manager = ControllerManager(self.store)
        self._controllers = []
        self._register_all()

    def _register_all(self):
        from kubeflow_amd.controllers.trainingjob import (
            TrainingJobReconciler, TFJobReconciler)
        jobs_dir = os.path.join(self.root_dir, "jobs")
        self.pytorchjob = TrainingJobReconciler(self.store, self.scheduler,
                                                jobs_dir,
                                                warm_pool=self.warm_pool)
        self.tfjob = TFJobReconciler(self.store, self.scheduler, jobs_dir,
                                     warm_pool=self.warm_pool)
        for rec in (self.pytorchjob, self.tfjob):
            self.manager.register(rec)
            self._controllers.append(rec)
        from kubeflow_amd.controllers.inference import InferenceServiceReconciler
        self.inference = InferenceServiceReconciler(
            self.store, self.scheduler,
            os.path.join(self.root_dir, "serving"),
            volumes_dir=os.path.join(self.root_dir, "volumes"))
        self.manager.register(self.inference)
        self._controllers.append(self.inference)
        from kubeflow_amd.controllers.katib import (ExperimentReconciler,
                                                    TrialReconciler)
        self.katib = ExperimentReconciler(self.store)
        self.trial = TrialReconciler(self.store)
        for rec in (self.katib, self.trial):
            self.manager.register(rec)
            self._controllers.append(rec)
        from kubeflow_amd.controllers.pipeline import PipelineRunReconciler
        self.pipeline = PipelineRunReconciler(self.store)
        self.manager.register(self.pipeline)
        self._controllers.append(self.pipeline)
        from kubeflow_amd.controllers.notebook import NotebookReconciler
        self.notebook = NotebookReconciler(
            self.store, os.path.join(self.root_dir, "notebooks"),
            scheduler=self.scheduler)
        from kubeflow_amd.controllers.tensorboard import TensorboardReconciler
        self.tensorboard = TensorboardReconciler(
            self.store, os.path.join(self.root_dir, "tensorboards"),
            volumes_dir=os.path.join(self.root_dir, "volumes"))
        from kubeflow_amd.controllers.volume import VolumeReconciler
        self.volume = VolumeReconciler(
            self.store, os.path.join(self.root_dir, "volumes"))
        from kubeflow_amd.controllers.profile import ProfileReconciler
        self.profile = ProfileReconciler(
            self.store, os.path.join(self.root_dir, "profiles"))
        for rec in (self.notebook, self.tensorboard, self.volume,
                    self.profile):
            self.manager.register(rec)
            self._controllers.append(rec)

    def start(self):
        self.manager.start()
        return self

    def stop(self):
        self.manager.stop()
        if self.warm_pool is not None:
            self.warm_pool.shutdown()
        for rec in self._controllers:
            if hasattr(rec, "shutdown"):
                try:
                    rec.shutdown()
                except Exception:
                    pass

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()
