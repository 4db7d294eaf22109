"""CI workflow runner: emit or submit per-component test workflows.

    python -m kubeflow_amd.ci.run --changed kubeflow_amd/api/server.py
    python -m kubeflow_amd.ci.run --changed ... --submit [--api URL]

Without --submit the selected PipelineRun manifests print as YAML (the
reference builders emit Argo YAML the same way); with --submit they POST
to a running platform API's pipelineruns route.
"""
from __future__ import annotations

import argparse
import json
import sys
import urllib.request

import yaml

from kubeflow_amd.ci import workflows_for_changes


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--changed", nargs="+", required=True)
    ap.add_argument("--namespace", default="kubeflow-ci")
    ap.add_argument("--submit", action="store_true")
    ap.add_argument("--api", default="http://127.0.0.1:8099")
    args = ap.parse_args()

    runs = workflows_for_changes(args.changed, namespace=args.namespace)
    if not runs:
        print("# no workflows triggered", file=sys.stderr)
        return 0
    if not args.submit:
        print(yaml.safe_dump_all(runs, sort_keys=False))
        return 0
    for run in runs:
        ns = run["metadata"]["namespace"]
        req = urllib.request.Request(
            f"{args.api}/api/namespaces/{ns}/pipelineruns",
            data=json.dumps(run).encode(),
            headers={"Content-Type": "application/json"}, method="POST")
        with urllib.request.urlopen(req, timeout=10) as r:
            body = json.load(r)
            print(f"submitted {run['metadata']['name']}: "
                  f"{body.get('message')}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
