"""CI workflow-builder layer (py/kubeflow/kubeflow/ci
builders + prow_config triggers, mapped onto PipelineRun + the worker's
pytest task kind)."""
import time

from kubeflow_amd.api import new_object
from kubeflow_amd.api.objects import has_condition
from kubeflow_amd.ci import TestBuilder, workflows_for_changes
from kubeflow_amd.platform import Platform


def test_trigger_routing():
    """prow_config semantics: changed dirs select exactly the matching
    component workflows."""
    runs = workflows_for_changes(["kubeflow_amd/api/server.py"])
    names = {r["metadata"]["labels"]["workflow"] for r in runs}
    assert names == {"central-dashboard"}
    runs = workflows_for_changes([
        "kubeflow_amd/scheduler/poddefaults.py",
        "kubeflow_amd/controllers/notebook.py"])
    names = {r["metadata"]["labels"]["workflow"] for r in runs}
    assert names == {"admission-webhook", "notebook-sessions"}
    assert workflows_for_changes(["README.md"]) == []


def test_builder_shape():
    run = TestBuilder("x", ["tests/test_config.py"]).build()
    assert run["kind"] == "PipelineRun"
    t = run["spec"]["tasks"][0]
    assert t["template"]["task"] == "pytest"
    assert t["template"]["pytest_args"] == ["tests/test_config.py"]
    assert run["metadata"]["labels"]["job-type"] == "presubmit"


def test_ci_workflow_executes(tmp_path):
    """A built workflow actually runs its pytest task through the
    platform's pipeline executor (the in-cluster-test half of the
    reference's Argo flow)."""
    with Platform(root_dir=str(tmp_path)) as plat:
        run = TestBuilder("smoke", ["tests/test_config.py"],
                          namespace="default").build()
        plat.store.create(run)
        deadline = time.time() + 240
        while time.time() < deadline:
            obj = plat.store.get("PipelineRun", "ci-smoke", "default")
            if has_condition(obj, "Succeeded") or has_condition(obj, "Failed"):
                break
            time.sleep(0.5)
        assert has_condition(obj, "Succeeded"), obj["status"]


def test_ci_workflow_reports_failure(tmp_path):
    """A failing selection fails the workflow (signal, not silence)."""
    with Platform(root_dir=str(tmp_path)) as plat:
        run = TestBuilder("bad", ["tests/no_such_test_file.py"],
                          namespace="default").build()
        plat.store.create(run)
        deadline = time.time() + 120
        while time.time() < deadline:
            obj = plat.store.get("PipelineRun", "ci-bad", "default")
            if has_condition(obj, "Succeeded") or has_condition(obj, "Failed"):
                break
            time.sleep(0.5)
        assert has_condition(obj, "Failed"), obj["status"]
