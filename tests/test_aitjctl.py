"""aitjctl CLI (cli.py) against the in-memory cluster."""
import io

import pytest

from trainingjob_operator_amd import cli
from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.kube.fake import FakeKubeApi

NS = "default"


@pytest.fixture
def cluster():
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    api.create_job(NS, {
        "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
        "metadata": {"name": "j", "namespace": NS},
        "spec": {"replicaSpecs": {"trainer": {
            "replicas": 2, "minReplicas": 1, "maxReplicas": 4,
            "edlPolicy": "Manual",
            "restartPolicy": "OnFailure", "restartScope": "All",
            "template": {"spec": {"containers": [{
                "name": "aitj-main",
                "ports": [{"name": "aitj-p", "containerPort": 5000}],
            }]}}}}},
    })
    tc.sync_once(f"{NS}/j")
    api.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/j")
    return api, tc


def test_get(cluster, capsys):
    api, _ = cluster
    assert cli.main(["get", "-n", NS], api=api) == 0
    out = capsys.readouterr().out
    assert "NAME" in out and "j" in out and "Running" in out
    assert "trainer:2/2" in out


def test_describe(cluster, capsys):
    api, _ = cluster
    assert cli.main(["describe", "j", "-n", NS], api=api) == 0
    out = capsys.readouterr().out
    assert "Phase:     Running" in out
    assert "trainer: 2 desired (min=1 max=4 edl=Manual)" in out
    assert "active=2" in out
    assert "Conditions:" in out


def test_resize_within_bounds(cluster, capsys):
    api, tc = cluster
    assert cli.main(["resize", "j", "4", "-n", NS], api=api) == 0
    assert api.get_job(NS, "j")["spec"]["replicaSpecs"]["trainer"][
        "replicas"] == 4
    # the controller picks it up as a world restart
    tc.sync_once(f"{NS}/j")
    from trainingjob_operator_amd.api.types import AITrainingJob
    assert AITrainingJob.from_dict(
        api.get_job(NS, "j")).status.restart_replica_name == "trainer"


def test_resize_rejects_out_of_bounds(cluster, capsys):
    api, _ = cluster
    assert cli.main(["resize", "j", "9", "-n", NS], api=api) == 1
    assert "maxReplicas" in capsys.readouterr().err
    assert cli.main(["resize", "j", "3", "--role", "nope", "-n", NS],
                    api=api) == 1


def test_delete(cluster, capsys):
    api, _ = cluster
    assert cli.main(["delete", "j", "-n", NS], api=api) == 0
    assert api.list_jobs(NS) == []


def test_logs(cluster, capsys):
    api, _ = cluster
    api.set_pod_log(NS, "j-trainer-0", "step 1 loss 6.2\nstep 2 loss 6.1")
    api.set_pod_log(NS, "j-trainer-1", "hello from rank 1")
    assert cli.main(["logs", "j", "-n", NS], api=api) == 0
    out = capsys.readouterr().out
    assert "==> j-trainer-0 <==" in out and "step 2 loss 6.1" in out
    assert "hello from rank 1" in out
    # single replica + tail
    assert cli.main(["logs", "j", "--replica", "trainer-0", "--tail", "1",
                     "-n", NS], api=api) == 0
    out = capsys.readouterr().out
    assert "step 2 loss 6.1" in out and "step 1" not in out
    assert "==>" not in out
    assert cli.main(["logs", "j", "--replica", "nope-9", "-n", NS],
                    api=api) == 1


def test_nodes(cluster, capsys, monkeypatch):
    import json
    from trainingjob_operator_amd.agent import gpu_health
    from trainingjob_operator_amd.agent.node_agent import NodeAgent
    api, _ = cluster
    api.add_node("gpu-node", ready=True)
    monkeypatch.setenv(gpu_health.FAKE_ENV, json.dumps(
        {"gpus": [{"index": 0, "temp_c": 60},
                  {"index": 1, "temp_c": 115, "message": "hot"}]}))
    NodeAgent(api, "gpu-node", expected_gpus=2).probe_and_publish()
    assert cli.main(["nodes"], api=api) == 0
    out = capsys.readouterr().out
    assert "NODE" in out and "gpu-node" in out
    assert "False" in out and "gpu1: hot" in out
    assert "node-0" in out   # plain node without the agent shows "-"


def test_describe_shows_events(cluster, capsys):
    api, _ = cluster
    assert cli.main(["describe", "j", "-n", NS], api=api) == 0
    out = capsys.readouterr().out
    assert "Events:" in out
    assert "SuccessfulCreatePod" in out


def test_get_all_namespaces(cluster, capsys):
    api, _ = cluster
    api.create_job("other", {
        "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
        "metadata": {"name": "elsewhere", "namespace": "other"},
        "spec": {"replicaSpecs": {"t": {"template": {"spec": {
            "containers": [{"name": "aitj-x"}]}}}}}})
    assert cli.main(["get", "-A"], api=api) == 0
    out = capsys.readouterr().out
    assert "j" in out and "elsewhere" in out
