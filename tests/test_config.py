"""Config tree + spawner readOnly semantics tests."""
import os

import pytest
import yaml

from kubeflow_amd import config as C


def test_defaults_and_yaml_merge(tmp_path, monkeypatch):
    f = tmp_path / "cfg.yaml"
    f.write_text(yaml.safe_dump({
        "scheduler": {"bucket_mb": 128},
        "spawner": {"image": {"value": "custom:1", "readOnly": True}},
    }))
    cfg = C.load(path=str(f), use_env=False, refresh=True)
    assert cfg["scheduler"]["bucket_mb"] == 128
    assert cfg["scheduler"]["gang_timeout_s"] == 600  # default preserved
    assert cfg["spawner"]["image"]["value"] == "custom:1"
    assert cfg["culler"]["idle_minutes"] == 1440


def test_env_override(monkeypatch):
    monkeypatch.setenv("KF_SCHEDULER_BUCKET_MB", "256")
    monkeypatch.setenv("KF_CULLER_ENABLED", "true")
    cfg = C.load(path="/nonexistent", refresh=True)
    assert cfg["scheduler"]["bucket_mb"] == 256
    assert cfg["culler"]["enabled"] is True
    C.load(path="/nonexistent", use_env=False, refresh=True)  # reset cache


def test_spawner_readonly_enforcement():
    cfg = C.load(path="/nonexistent", use_env=False, refresh=True)
    req = {"image": "evil:latest", "cpu": "8", "shm": False}
    out = C.enforce_spawner(cfg, req)
    assert out["image"] == "evil:latest"   # writable -> user wins
    assert out["cpu"] == "8"
    assert out["shm"] is True              # readOnly -> admin wins
