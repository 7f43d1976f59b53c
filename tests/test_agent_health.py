"""Node agent: FAKE_GPU health probing and the full health -> NodeFail ->
restart path through the controller."""
import json

import pytest

from trainingjob_operator_amd.agent import gpu_health
from trainingjob_operator_amd.agent.node_agent import (
    GPU_HEALTH_ANNOTATION, GPU_HEALTH_CONDITION, NodeAgent,
)
from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.types import AITrainingJob, Phase
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.kube.fake import FakeKubeApi

NS = "default"


def fake_health(monkeypatch, payload):
    monkeypatch.setenv(gpu_health.FAKE_ENV, json.dumps(payload))


def test_probe_fake_healthy(monkeypatch):
    fake_health(monkeypatch, {"expected": 8, "gpus": [
        {"index": i, "temp_c": 60} for i in range(8)]})
    report = gpu_health.probe()
    assert report.healthy
    assert "8 GPUs healthy" in report.summary()


@pytest.mark.parametrize("payload,expect_msg", [
    ({"expected": 8, "gpus": [{"index": i} for i in range(7)]}, "GPU lost"),
    ({"gpus": [{"index": 0, "ecc_uncorrectable": 3,
                "message": "ECC"}]}, "gpu0"),
    ({"gpus": [{"index": 0, "temp_c": 115, "message": "hot"}]}, "gpu0"),
    ({"gpus": [{"index": 0, "xgmi_ok": False,
                "message": "xGMI link down"}]}, "xGMI"),
    ({"gpus": [], "probe_error": "rocm-smi hung"}, "probe error"),
])
def test_probe_fake_unhealthy(monkeypatch, payload, expect_msg):
    fake_health(monkeypatch, payload)
    report = gpu_health.probe()
    assert not report.healthy
    assert expect_msg in report.summary()


def test_agent_publishes_condition(monkeypatch):
    api = FakeKubeApi()
    api.add_node("gpu-node", ready=True)
    fake_health(monkeypatch, {"gpus": [{"index": 0, "temp_c": 50}]})
    agent = NodeAgent(api, "gpu-node", expected_gpus=1)
    agent.probe_and_publish()
    node = api.get_node("gpu-node")
    conds = {c["type"]: c for c in node["status"]["conditions"]}
    assert conds[GPU_HEALTH_CONDITION]["status"] == "True"
    detail = json.loads(
        node["metadata"]["annotations"][GPU_HEALTH_ANNOTATION])
    assert detail["healthy"]

    # flip to GPU-lost
    fake_health(monkeypatch, {"expected": 1, "gpus": []})
    agent.probe_and_publish()
    node = api.get_node("gpu-node")
    conds = {c["type"]: c for c in node["status"]["conditions"]}
    assert conds[GPU_HEALTH_CONDITION]["status"] == "False"
    assert "GPU lost" in conds[GPU_HEALTH_CONDITION]["message"]
    # kubelet Ready untouched
    assert conds["Ready"]["status"] == "True"


def test_gpu_unhealthy_drives_nodefail_restart(monkeypatch):
    """End-to-end: agent flags the GPU -> controller NodeFail -> restart."""
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    api.create_job(NS, {
        "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
        "metadata": {"name": "j", "namespace": NS},
        "spec": {"replicaSpecs": {"trainer": {
            "replicas": 2, "restartPolicy": "OnNodeFail",
            "restartScope": "All", "restartLimit": 2,
            "template": {"spec": {"containers": [{
                "name": "aitj-main",
                "ports": [{"name": "aitj-p", "containerPort": 5000}],
            }]}}}}},
    })
    tc.sync_once(f"{NS}/j")
    api.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/j")
    job = AITrainingJob.from_dict(api.get_job(NS, "j"))
    assert job.status.phase == Phase.RUNNING

    # the node agent reports an uncorrectable ECC error on node-0
    fake_health(monkeypatch, {"gpus": [
        {"index": 0, "ecc_uncorrectable": 2, "message": "UE ECC"}]})
    NodeAgent(api, "node-0").probe_and_publish()

    tc.sync_once(f"{NS}/j")
    job = AITrainingJob.from_dict(api.get_job(NS, "j"))
    assert job.status.restart_replica_name == "trainer"
    assert api.pod_names(NS) == []  # force-evicted
    # heal the node so the restarted pods land on a healthy one
    fake_health(monkeypatch, {"gpus": [{"index": 0}]})
    NodeAgent(api, "node-0").probe_and_publish()
    tc.sync_once(f"{NS}/j")
    assert AITrainingJob.from_dict(
        api.get_job(NS, "j")).status.phase == Phase.RESTARTING
    tc.sync_once(f"{NS}/j")
    assert len(api.pod_names(NS)) == 2


def test_probe_amd_smi_parsing(monkeypatch):
    from trainingjob_operator_amd.agent import gpu_health as gh
    monkeypatch.delenv(gh.FAKE_ENV, raising=False)

    def fake_run_json(cmd, timeout=10.0):
        if cmd[0] == "amd-smi":
            return [
                {"gpu": 0, "temperature": {"edge": {"value": 55}},
                 "ecc": {"total_uncorrectable_count": 0}},
                {"gpu": 1, "temperature": {"edge": {"value": 90}},
                 "ecc": {"total_uncorrectable_count": 2}},
            ]
        return None

    monkeypatch.setattr(gh, "_run_json", fake_run_json)
    report = gh.probe(expected=2)
    assert len(report.gpus) == 2
    assert report.gpus[0].healthy
    assert not report.gpus[1].healthy
    assert "uncorrectable" in report.gpus[1].message
    assert not report.healthy


def test_probe_rocm_smi_fallback(monkeypatch):
    from trainingjob_operator_amd.agent import gpu_health as gh
    monkeypatch.delenv(gh.FAKE_ENV, raising=False)

    def fake_run_json(cmd, timeout=10.0):
        if cmd[0] == "amd-smi":
            return None  # not installed
        if cmd[0] == "rocm-smi":
            return {"card0": {"Temperature (Sensor edge) (C)": "62.0"},
                    "card1": {"Temperature (Sensor junction) (C)": "115.0"}}
        return None

    monkeypatch.setattr(gh, "_run_json", fake_run_json)
    report = gh.probe()
    assert len(report.gpus) == 2
    assert report.gpus[0].healthy
    assert not report.gpus[1].healthy  # 115C >= limit
