"""CI workflow builders — the reference's per-component workflow-authoring
layer (py/kubeflow/kubeflow/ci/*) mapped onto the
platform's own PipelineRun executor:

  * `TestBuilder` plays `ArgoTestBuilder` (ci/workflow_utils.py:31): it
    authors a PipelineRun whose tasks are pytest runs executed by the
    worker runtime's `task: pytest` kind (no Argo, no Kaniko — images are
    moot in the process model, so each builder covers the unit-test half
    of its reference workflow);
  * `TRIGGERS` plays prow_config.yaml:8-40: path globs -> builder, so
    `workflows_for_changes(paths)` returns exactly the per-component
    workflows a presubmit for those paths would run;
  * `python -m kubeflow_amd.ci.run --changed a.py b.py [--submit]` emits
    the selected PipelineRun manifests (YAML) or submits them to a
    running platform API.
"""
from __future__ import annotations

import fnmatch
from typing import Dict, List, Optional

from kubeflow_amd.api import new_object


class TestBuilder:
    """Authors one component's test workflow (ArgoTestBuilder analog)."""

    def __init__(self, name: str, tests: List[str],
                 namespace: str = "kubeflow-ci",
                 extra_tasks: Optional[List[dict]] = None):
        self.name = name
        self.tests = tests
        self.namespace = namespace
        self.extra_tasks = extra_tasks or []

    def build(self) -> dict:
        """Build the PipelineRun (the reference returns an Argo Workflow
        dict; same role)."""
        tasks = [{
            "name": f"pytest-{i}",
            "dependencies": [],
            "template": {"task": "pytest", "pytest_args": [t],
                         "gpus_per_replica": 0},
        } for i, t in enumerate(self.tests)]
        for t in self.extra_tasks:
            tasks.append(t)
        run = new_object("PipelineRun", f"ci-{self.name}", self.namespace,
                         spec={"tasks": tasks},
                         api_version="pipelines.kubeflow.org/v1")
        run["metadata"]["labels"]["workflow"] = self.name
        run["metadata"]["labels"]["job-type"] = "presubmit"
        return run


# prow_config.yaml analog: include_dirs globs -> builder factory.
# Each entry mirrors a reference workflow (named in the comment).
TRIGGERS: List[dict] = [
    {   # notebook_controller_tests.py + jwa_tests
        "name": "notebook-sessions",
        "include_dirs": ["kubeflow_amd/controllers/notebook.py",
                         "kubeflow_amd/runtime/notebook_server.py",
                         "kubeflow_amd/controllers/tensorboard.py",
                         "kubeflow_amd/controllers/volume.py"],
        "tests": ["tests/test_sessions.py"],
    },
    {   # access_management_tests.py + profile controller
        "name": "access-management",
        "include_dirs": ["kubeflow_amd/kfam/*",
                         "kubeflow_amd/controllers/profile.py"],
        "tests": ["tests/test_api_server.py",
                  "tests/test_store_controllers.py"],
    },
    {   # admission_webhook_tests.py
        "name": "admission-webhook",
        "include_dirs": ["kubeflow_amd/scheduler/poddefaults.py",
                         "kubeflow_amd/scheduler/launcher.py"],
        "tests": ["tests/test_poddefaults.py", "tests/test_scheduler.py"],
    },
    {   # central_dashboard_tests.py + common_ui_tests.py
        "name": "central-dashboard",
        "include_dirs": ["kubeflow_amd/api/*", "kubeflow_amd/dashboard/*"],
        "tests": ["tests/test_api_server.py", "tests/test_dashboard.py"],
    },
    {   # training / operators (sibling-repo integration promoted in-repo)
        "name": "training-operator",
        "include_dirs": ["kubeflow_amd/controllers/trainingjob.py",
                         "kubeflow_amd/runtime/worker.py",
                         "kubeflow_amd/runtime/trainer.py",
                         "kubeflow_amd/parallel/*"],
        "tests": ["tests/test_pytorchjob_e2e.py", "tests/test_ddp_gloo.py",
                  "tests/test_flat_and_trainer.py"],
    },
    {   # serving
        "name": "kserve",
        "include_dirs": ["kubeflow_amd/controllers/inference.py",
                         "kubeflow_amd/runtime/serving*.py"],
        "tests": ["tests/test_serving.py"],
    },
    {   # katib + pipelines
        "name": "katib-pipelines",
        "include_dirs": ["kubeflow_amd/controllers/katib.py",
                         "kubeflow_amd/controllers/pipeline.py",
                         "kubeflow_amd/katib/*"],
        "tests": ["tests/test_katib_pipeline.py"],
    },
    {   # kernels / ops (no reference analog — the compute path's CI)
        "name": "ops-kernels",
        "include_dirs": ["kubeflow_amd/ops/*"],
        "tests": ["tests/test_ops_reference.py"],
    },
]


def _matches(path: str, patterns: List[str]) -> bool:
    return any(fnmatch.fnmatch(path, pat) or
               fnmatch.fnmatch(path, pat + "*") or
               path.startswith(pat.rstrip("*"))
               for pat in patterns)


def workflows_for_changes(changed: List[str],
                          namespace: str = "kubeflow-ci") -> List[dict]:
    """Presubmit routing (prow_config semantics): every trigger whose
    include_dirs match a changed path contributes its workflow once."""
    out = []
    for trig in TRIGGERS:
        if any(_matches(p, trig["include_dirs"]) for p in changed):
            out.append(TestBuilder(trig["name"], trig["tests"],
                                   namespace=namespace).build())
    return out
