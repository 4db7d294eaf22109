"""Tensorboard-equivalent metrics viewer process.

The reference's tensorboard-controller deploys `tensorflow/tensorflow:2.1.0
tensorboard --logdir` (tensorboard_controller.go:152); this offline image
has no TensorBoard, so the session process serves the same role over the
job metrics this platform actually produces: it scans the logdir for
status.json / metrics.jsonl files and serves scalar series as JSON.

    GET /healthz
    GET /data/runs                 list of run names found under logdir
    GET /data/scalars?run=<name>   scalar series for a run
"""
from __future__ import annotations

import argparse
import glob
import json
import os

from fastapi import FastAPI
import uvicorn


def scan_runs(logdir: str):
    runs = {}
    for path in glob.glob(os.path.join(logdir, "**", "status.json"),
                          recursive=True):
        run = os.path.relpath(os.path.dirname(path), logdir)
        try:
            with open(path) as f:
                data = json.load(f)
            runs.setdefault(run, []).append(
                {"step": data.get("step"), "metrics": data.get("metrics")})
        except (OSError, json.JSONDecodeError):
            continue
    for path in glob.glob(os.path.join(logdir, "**", "metrics.jsonl"),
                          recursive=True):
        run = os.path.relpath(os.path.dirname(path), logdir)
        series = []
        try:
            with open(path) as f:
                for line in f:
                    if not line.strip():
                        continue
                    try:
                        series.append(json.loads(line))
                    except json.JSONDecodeError:
                        continue  # torn concurrent append: skip the line
        except OSError:
            continue
        runs.setdefault(run, []).extend(series)
    return runs


def build_app(logdir: str) -> FastAPI:
    app = FastAPI(title="kubeflow-amd tensorboard")

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "logdir": logdir}

    @app.get("/data/runs")
    def data_runs():
        return {"runs": sorted(scan_runs(logdir).keys())}

    @app.get("/data/scalars")
    def data_scalars(run: str = ""):
        runs = scan_runs(logdir)
        if run:
            return {"run": run, "points": runs.get(run, [])}
        return runs

    return app


def main(argv=None):
    ap = argparse.ArgumentParser()
    ap.add_argument("--spec", required=True)
    args = ap.parse_args(argv)
    with open(args.spec) as f:
        spec = json.load(f)
    app = build_app(spec.get("logdir", "."))
    uvicorn.run(app, host="127.0.0.1", port=int(spec.get("port", 6006)),
                log_level="warning")


if __name__ == "__main__":
    main()
