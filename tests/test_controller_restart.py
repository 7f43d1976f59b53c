"""The two-sync restart dance and restart scope semantics:
Terminating -> (pods gone) -> Restarting -> recreate with bumped
RestartCount label/env, restart limits, node-fail force deletion."""
import pytest

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.types import AITrainingJob, Phase
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.kube.fake import FakeKubeApi

NS = "default"


def make_job(name="job", scope="All", policy="ExitCode", limit=3,
             second_role=False):
    specs = {
        "trainer": {
            "replicas": 2,
            "restartPolicy": policy,
            "restartScope": scope,
            "restartLimit": limit,
            "template": {"spec": {"containers": [{
                "name": "aitj-main",
                "ports": [{"name": "aitj-p", "containerPort": 5000}],
            }]}},
        }
    }
    if second_role:
        specs["pserver"] = {
            "replicas": 1,
            "restartPolicy": "Never",
            "template": {"spec": {"containers": [{
                "name": "aitj-ps",
                "ports": [{"name": "aitj-p", "containerPort": 6000}],
            }]}},
        }
    return {
        "apiVersion": C.API_VERSION,
        "kind": C.CRD_KIND,
        "metadata": {"name": name, "namespace": NS},
        "spec": {"restartingExitCode": "137,128", "replicaSpecs": specs},
    }


@pytest.fixture
def cluster():
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    return api, tc


def sync(tc, name="job", times=1):
    for _ in range(times):
        tc.sync_once(f"{NS}/{name}")


def job_of(api, name="job"):
    return AITrainingJob.from_dict(api.get_job(NS, name))


def start_running(api, tc, name="job"):
    sync(tc, name)
    api.set_all_pods_phase(NS, "Running")
    sync(tc, name)
    assert job_of(api, name).status.phase == Phase.RUNNING


def test_restart_scope_all_full_dance(cluster):
    api, tc = cluster
    api.create_job(NS, make_job(scope="All", second_role=True))
    start_running(api, tc)
    assert len(api.pod_names(NS)) == 3

    # retryable failure on trainer-1
    api.set_pod_phase(NS, "job-trainer-1", "Failed", exit_code=137)
    sync(tc)
    job = job_of(api)
    # sync A: ALL pods (both roles) deleted, marker set, phase Terminating
    assert api.pod_names(NS) == []
    assert job.status.phase == Phase.TERMINATING
    assert job.status.restart_replica_name == "trainer"
    # scope All bumps every role's counter (status.go:322-330)
    assert job.status.restart_counts == {"trainer": 1, "pserver": 1}

    # sync B: pods gone -> Restarting, marker cleared
    sync(tc)
    job = job_of(api)
    assert job.status.phase == Phase.RESTARTING
    assert job.status.restart_replica_name == ""

    # sync C: pods recreated with bumped RestartCount
    sync(tc)
    assert len(api.pod_names(NS)) == 3
    pod = api.get_pod(NS, "job-trainer-1")
    assert pod["metadata"]["labels"]["RestartCount"] == "1"
    env = {e["name"]: e["value"]
           for e in pod["spec"]["containers"][0]["env"]}
    assert env["TRAININGJOB_REPLICA_RESTARTCOUNT"] == "1"

    # back to Running
    api.set_all_pods_phase(NS, "Running")
    sync(tc)
    assert job_of(api).status.phase == Phase.RUNNING


def test_restart_scope_pod_only_deletes_one(cluster):
    api, tc = cluster
    api.create_job(NS, make_job(scope="Pod", second_role=True))
    start_running(api, tc)
    api.set_pod_phase(NS, "job-trainer-1", "Failed", exit_code=137)
    sync(tc)
    # only the failed pod deleted
    assert api.pod_names(NS) == ["job-pserver-0", "job-trainer-0"]
    job = job_of(api)
    assert job.status.restart_counts["trainer"] == 1
    assert job.status.restart_counts.get("pserver", 0) == 0
    sync(tc)  # wait-gate: replica pods < replicas -> Restarting
    assert job_of(api).status.phase == Phase.RESTARTING
    sync(tc)
    assert len(api.pod_names(NS)) == 3


def test_restart_scope_replica_deletes_role(cluster):
    api, tc = cluster
    api.create_job(NS, make_job(scope="Replica", second_role=True))
    start_running(api, tc)
    api.set_pod_phase(NS, "job-trainer-0", "Failed", exit_code=128)
    sync(tc)
    assert api.pod_names(NS) == ["job-pserver-0"]
    sync(tc)
    assert job_of(api).status.phase == Phase.RESTARTING
    sync(tc)
    assert len(api.pod_names(NS)) == 3


def test_non_retryable_exit_code_fails_job(cluster):
    api, tc = cluster
    api.create_job(NS, make_job(policy="ExitCode"))
    start_running(api, tc)
    api.set_pod_phase(NS, "job-trainer-0", "Failed", exit_code=1)
    sync(tc)  # 1 not in "137,128" -> no restart; failPolicy Any -> terminate
    job = job_of(api)
    assert job.status.phase == Phase.TERMINATING
    assert Phase.FAILED in job.annotations
    sync(tc)
    assert job_of(api).status.phase == Phase.FAILED


def test_restart_limit_exhaustion(cluster):
    api, tc = cluster
    api.create_job(NS, make_job(scope="All", limit=1))
    start_running(api, tc)
    # first failure: restart allowed
    api.set_pod_phase(NS, "job-trainer-0", "Failed", exit_code=137)
    sync(tc, times=3)
    assert job_of(api).status.restart_counts["trainer"] == 1
    api.set_all_pods_phase(NS, "Running")
    sync(tc)
    # second failure: limit reached -> job fails
    api.set_pod_phase(NS, "job-trainer-0", "Failed", exit_code=137)
    sync(tc, times=2)
    assert job_of(api).status.phase in (Phase.TERMINATING, Phase.FAILED)
    sync(tc)
    assert job_of(api).status.phase == Phase.FAILED


def test_node_fail_restart(cluster):
    api, tc = cluster
    job = make_job(policy="OnNodeFail", scope="All")
    api.create_job(NS, job)
    start_running(api, tc)
    # node dies
    api.set_node_ready("node-0", False)
    api.add_node("node-1", ready=True)
    api.default_node = "node-1"
    sync(tc)
    j = job_of(api)
    assert j.status.restart_replica_name == "trainer"
    # force (grace 0) deletion was used for node-fail
    deletes = [a for a in api.actions if a[0] == "delete" and a[1] == "pod"]
    assert all(a[4] == 0 for a in deletes)
    sync(tc)
    assert job_of(api).status.phase == Phase.RESTARTING
    sync(tc)
    assert len(api.pod_names(NS)) == 2  # recreated (on the healthy node)


def test_node_fail_without_policy_fails_job(cluster):
    api, tc = cluster
    api.create_job(NS, make_job(policy="Never"))
    start_running(api, tc)
    api.set_node_ready("node-0", False)
    sync(tc)
    j = job_of(api)
    assert j.status.phase == Phase.TERMINATING
    assert Phase.NODE_FAIL in j.annotations
    sync(tc)
    assert job_of(api).status.phase == Phase.NODE_FAIL
