"""Tracing subsystem: JSONL events from controller syncs and spans."""
import json

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.kube.fake import FakeKubeApi
from trainingjob_operator_amd.utils.tracing import Tracer


def test_tracer_writes_jsonl(tmp_path):
    path = tmp_path / "trace.jsonl"
    t = Tracer(str(path), component="test")
    t.event("hello", a=1)
    with t.span("work", job="x"):
        pass
    lines = [json.loads(l) for l in path.read_text().splitlines()]
    assert lines[0]["kind"] == "hello" and lines[0]["a"] == 1
    assert lines[1]["kind"] == "work" and lines[1]["ok"]
    assert "duration_ms" in lines[1]


def test_tracer_disabled_is_noop():
    t = Tracer(None)
    assert not t.enabled
    t.event("x")  # no crash


def test_controller_emits_sync_and_transition_events(tmp_path, monkeypatch):
    path = tmp_path / "ctl.jsonl"
    import trainingjob_operator_amd.utils.tracing as tracing
    monkeypatch.setattr(tracing, "_global", None)
    monkeypatch.setenv("AITJ_TRACE", str(path))
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    api.create_job("default", {
        "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
        "metadata": {"name": "t", "namespace": "default"},
        "spec": {"replicaSpecs": {"trainer": {
            "replicas": 1,
            "template": {"spec": {"containers": [{"name": "aitj-m"}]}}}}},
    })
    tc.sync_once("default/t")
    api.set_all_pods_phase("default", "Running")
    tc.sync_once("default/t")
    events = [json.loads(l) for l in path.read_text().splitlines()]
    kinds = [e["kind"] for e in events]
    assert "sync" in kinds
    assert any(e["kind"] == "phase_transition" and e["to"] == "Running"
               for e in events)
    monkeypatch.setattr(tracing, "_global", None)
