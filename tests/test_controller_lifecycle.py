"""Reconcile-loop lifecycle tests against the in-memory fake API
(SURVEY.md §4 test plan item 2): create job -> pods + headless services with
exact labels/env; drive pod phases -> job phase machine; completion policies.
"""
import pytest

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.types import AITrainingJob, Phase
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.kube.fake import FakeKubeApi

NS = "default"


def make_job(name="mnist", replicas=2, rtype="trainer", **spec_over):
    spec = {
        "restartingExitCode": "137,128",
        "replicaSpecs": {
            rtype: {
                "replicas": replicas,
                "restartPolicy": "ExitCode",
                "restartScope": "All",
                "restartLimit": 3,
                "template": {"spec": {"containers": [{
                    "name": "aitj-main",
                    "image": "pytorch-rocm:latest",
                    "ports": [{"name": "aitj-port", "containerPort": 23456}],
                    "resources": {"limits": {"amd.com/gpu": 1}},
                }]}},
            }
        },
    }
    spec.update(spec_over)
    return {
        "apiVersion": C.API_VERSION,
        "kind": C.CRD_KIND,
        "metadata": {"name": name, "namespace": NS},
        "spec": spec,
    }


@pytest.fixture
def cluster():
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    return api, tc


def sync(tc, name="mnist", times=1):
    for _ in range(times):
        tc.sync_once(f"{NS}/{name}")


def job_of(api, name="mnist"):
    return AITrainingJob.from_dict(api.get_job(NS, name))


def test_create_pods_and_services(cluster):
    api, tc = cluster
    api.create_job(NS, make_job())
    sync(tc)
    assert api.pod_names(NS) == ["mnist-trainer-0", "mnist-trainer-1"]
    assert api.service_names(NS) == ["mnist-trainer-0", "mnist-trainer-1"]

    pod = api.get_pod(NS, "mnist-trainer-1")
    labels = pod["metadata"]["labels"]
    assert labels[C.LABEL_GROUP_NAME] == "elasticdeeplearning.ai"
    assert labels[C.LABEL_JOB_NAME] == "mnist"
    assert labels["JobName"] == "mnist"
    assert labels["PodRole"] == "trainer"
    assert labels["RestartCount"] == "0"
    assert labels[C.LABEL_REPLICA_NAME] == "trainer"
    assert labels[C.LABEL_REPLICA_INDEX] == "1"
    ref = pod["metadata"]["ownerReferences"][0]
    assert ref["kind"] == "AITrainingJob" and ref["controller"]
    assert pod["spec"]["restartPolicy"] == "Never"

    svc = api.services[(NS, "mnist-trainer-0")]
    assert svc["spec"]["clusterIP"] == "None"
    assert svc["spec"]["selector"][C.LABEL_REPLICA_INDEX] == "0"
    assert svc["spec"]["ports"] == [{"name": "aitj-23456", "port": 23456}]


def test_env_contract(cluster):
    api, tc = cluster
    api.create_job(NS, make_job())
    sync(tc)
    pod = api.get_pod(NS, "mnist-trainer-1")
    env = {e["name"]: e["value"]
           for e in pod["spec"]["containers"][0]["env"]}
    # reference contract (pod.go:548-652)
    assert env["TRAINER_INSTANCES"] == \
        "mnist-trainer-0.default,mnist-trainer-1.default"
    assert env["TRAINER_INSTANCES_NUM"] == "2"
    assert env["TRAINER_PORTS"] == "23456"
    assert env["TRAINER_PORTS_NUM"] == "1"
    assert env["TRAINER_HOSTS"] == \
        "mnist-trainer-0.default:23456,mnist-trainer-1.default:23456"
    assert env["TRAINER_HOSTS_NUM"] == "2"
    assert env["TRAININGJOB_REPLICA_NAME"] == "trainer"
    assert env["TRAININGJOB_REPLICA_INDEX"] == "1"
    assert env["TRAININGJOB_REPLICA_RESTARTCOUNT"] == "0"
    assert env["TRAININGJOB_SERVICE"] == "mnist-trainer-1.default"
    assert env["TRAININGJOB_NAME"] == "mnist"
    assert env["TRAININGJOB_NAMESPACE"] == "default"
    assert env["TRAININGJOB_PORTS"] == "23456"
    # MI355X RCCL extension
    assert env["MASTER_ADDR"] == "mnist-trainer-0.default"
    assert env["MASTER_PORT"] == "23456"
    assert env["WORLD_SIZE"] == "2"
    assert env["RANK"] == "1"
    assert env["LOCAL_RANK"] == "0"
    assert env["TRAININGJOB_RENDEZVOUS_EPOCH"] == "0"


def test_phase_progression_to_running(cluster):
    api, tc = cluster
    api.create_job(NS, make_job())
    sync(tc)
    # pods scheduled (auto bound) but containers creating
    api.set_all_pods_phase(NS, "Pending")
    sync(tc, times=2)
    assert job_of(api).status.phase == Phase.CREATING
    api.set_all_pods_phase(NS, "Running")
    sync(tc)
    job = job_of(api)
    assert job.status.phase == Phase.RUNNING
    assert job.status.start_running_time is not None
    assert job.status.replica_statuses["trainer"].active == 2


def test_complete_policy_all(cluster):
    api, tc = cluster
    api.create_job(NS, make_job())
    sync(tc)
    api.set_all_pods_phase(NS, "Running")
    sync(tc)
    api.set_pod_phase(NS, "mnist-trainer-0", "Succeeded")
    sync(tc)
    assert job_of(api).status.phase == Phase.RUNNING  # default All waits
    api.set_pod_phase(NS, "mnist-trainer-1", "Succeeded")
    sync(tc)
    job = job_of(api)
    # terminateTrainingJob: pods deleted, Terminating with annotation
    assert job.status.phase == Phase.TERMINATING
    assert Phase.SUCCEEDED in job.annotations
    assert api.pod_names(NS) == []
    assert api.service_names(NS) == []
    sync(tc)  # deferred finalization: pods gone -> final phase
    job = job_of(api)
    assert job.status.phase == Phase.SUCCEEDED
    assert job.status.end_time is not None
    # conditions audit trail: last is Succeed=True, previous flipped False
    assert job.status.conditions[-1].type == Phase.SUCCEEDED
    assert job.status.conditions[-1].status == "True"
    assert all(c.status == "False" for c in job.status.conditions[:-1])


def test_fail_policy_any_terminal(cluster):
    api, tc = cluster
    job = make_job()
    job["spec"]["replicaSpecs"]["trainer"]["restartPolicy"] = "Never"
    api.create_job(NS, job)
    sync(tc)
    api.set_all_pods_phase(NS, "Running")
    sync(tc)
    api.set_pod_phase(NS, "mnist-trainer-1", "Failed", exit_code=1)
    sync(tc)
    j = job_of(api)
    assert j.status.phase == Phase.TERMINATING
    assert Phase.FAILED in j.annotations
    sync(tc)
    assert job_of(api).status.phase == Phase.FAILED


def test_clean_pod_policy_none_keeps_pods(cluster):
    api, tc = cluster
    api.create_job(NS, make_job(cleanPodPolicy="None"))
    sync(tc)
    api.set_all_pods_phase(NS, "Succeeded")
    sync(tc)
    job = job_of(api)
    assert job.status.phase == Phase.SUCCEEDED
    assert len(api.pod_names(NS)) == 2  # kept


def test_completed_job_not_resynced(cluster):
    api, tc = cluster
    api.create_job(NS, make_job(cleanPodPolicy="None"))
    sync(tc)
    api.set_all_pods_phase(NS, "Succeeded")
    sync(tc, times=3)
    job = job_of(api)
    assert job.status.phase == Phase.SUCCEEDED
    n_conditions = len(job.status.conditions)
    sync(tc, times=2)  # terminal phase gate: no further mutations
    assert len(job_of(api).status.conditions) == n_conditions


def test_gap_fill_recreates_missing_pod(cluster):
    api, tc = cluster
    api.create_job(NS, make_job())
    sync(tc)
    api.delete_pod(NS, "mnist-trainer-0")
    sync(tc)
    assert "mnist-trainer-0" in api.pod_names(NS)


def test_rank0_complete_policy(cluster):
    api, tc = cluster
    job = make_job()
    job["spec"]["replicaSpecs"]["trainer"]["completePolicy"] = "Rank0"
    api.create_job(NS, job)
    sync(tc)
    api.set_all_pods_phase(NS, "Running")
    sync(tc)
    # rank 1 succeeding does nothing under Rank0
    api.set_pod_phase(NS, "mnist-trainer-1", "Succeeded")
    sync(tc)
    assert job_of(api).status.phase == Phase.RUNNING
    api.set_pod_phase(NS, "mnist-trainer-0", "Succeeded")
    sync(tc, times=2)
    assert job_of(api).status.phase == Phase.SUCCEEDED


def test_validation_rejected_job_gets_event(cluster):
    api, tc = cluster
    bad = make_job()
    bad["spec"]["replicaSpecs"]["trainer"]["restartPolicy"] = "Nope"
    api.create_job(NS, bad)
    sync(tc)
    assert api.pod_names(NS) == []
    assert any(e["reason"] == "ValidationFailed" for e in api.events)


def test_timelimit_timeout(cluster):
    api, tc = cluster
    import time
    api.create_job(NS, make_job(timeLimit=100))
    now = time.time()
    tc.sync_once(f"{NS}/mnist", now=now)
    api.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/mnist", now=now + 1)
    assert job_of(api).status.phase == Phase.RUNNING
    tc.sync_once(f"{NS}/mnist", now=now + 200)
    j = job_of(api)
    assert j.status.phase == Phase.TERMINATING
    assert Phase.TIMEOUT in j.annotations
    tc.sync_once(f"{NS}/mnist", now=now + 201)
    assert job_of(api).status.phase == Phase.TIMEOUT


def test_preempted_annotation(cluster):
    api, tc = cluster
    api.create_job(NS, make_job())
    sync(tc)
    api.set_all_pods_phase(NS, "Running")
    sync(tc)
    # external preemption signal (reference: pod.go:160-162)
    j = api.get_job(NS, "mnist")
    j["metadata"].setdefault("annotations", {})[Phase.PREEMPTED] = \
        "preempted by scheduler"
    api.update_job(NS, "mnist", j)
    sync(tc, times=2)
    assert job_of(api).status.phase in (Phase.TERMINATING, Phase.PREEMPTED)
    sync(tc)
    assert job_of(api).status.phase == Phase.PREEMPTED
