"""Elastic resize semantics (real semantics for minReplicas/maxReplicas/
edlPolicy, which the reference declared but never read — SURVEY.md §C15):
mutating spec.replicas within [min,max] triggers a coordinated world restart
with a bumped rendezvous epoch; out-of-range pods are cleaned up."""
import pytest

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.types import AITrainingJob, Phase
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.controller.pods import EPOCH_ANNOTATION
from trainingjob_operator_amd.kube.fake import FakeKubeApi

NS = "default"


def make_job(replicas=2):
    return {
        "apiVersion": C.API_VERSION,
        "kind": C.CRD_KIND,
        "metadata": {"name": "elastic", "namespace": NS},
        "spec": {"replicaSpecs": {"trainer": {
            "replicas": replicas,
            "minReplicas": 2,
            "maxReplicas": 8,
            "edlPolicy": "Manual",
            "restartPolicy": "OnFailure",
            "restartScope": "All",
            "template": {"spec": {"containers": [{
                "name": "aitj-main",
                "ports": [{"name": "aitj-p", "containerPort": 5000}],
            }]}},
        }}},
    }


@pytest.fixture
def cluster():
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    return api, tc


def sync(tc, times=1):
    for _ in range(times):
        tc.sync_once(f"{NS}/elastic")


def job_of(api):
    return AITrainingJob.from_dict(api.get_job(NS, "elastic"))


def resize(api, n):
    j = api.get_job(NS, "elastic")
    j["spec"]["replicaSpecs"]["trainer"]["replicas"] = n
    api.update_job(NS, "elastic", j)


def test_scale_up_bumps_epoch_and_world(cluster):
    api, tc = cluster
    api.create_job(NS, make_job(replicas=2))
    sync(tc)
    api.set_all_pods_phase(NS, "Running")
    sync(tc)
    assert len(api.pod_names(NS)) == 2

    resize(api, 8)
    sync(tc)  # stale world detected -> all role pods deleted, epoch bumped
    j = job_of(api)
    assert j.annotations[EPOCH_ANNOTATION] == "1"
    assert j.status.restart_replica_name == "trainer"
    assert api.pod_names(NS) == []
    sync(tc)  # wait-gate -> Restarting
    assert job_of(api).status.phase == Phase.RESTARTING
    sync(tc)  # recreate at new world size
    assert len(api.pod_names(NS)) == 8
    pod = api.get_pod(NS, "elastic-trainer-7")
    env = {e["name"]: e["value"]
           for e in pod["spec"]["containers"][0]["env"]}
    assert env["WORLD_SIZE"] == "8"
    assert env["TRAININGJOB_RENDEZVOUS_EPOCH"] == "1"
    assert env["TRAININGJOB_MIN_REPLICAS"] == "2"
    assert env["TRAININGJOB_MAX_REPLICAS"] == "8"
    # elastic resize must NOT consume the restart budget
    assert job_of(api).status.restart_counts.get("trainer", 0) == 0


def test_scale_down_cleans_out_of_range(cluster):
    api, tc = cluster
    api.create_job(NS, make_job(replicas=4))
    sync(tc)
    api.set_all_pods_phase(NS, "Running")
    sync(tc)
    assert len(api.pod_names(NS)) == 4
    resize(api, 2)
    sync(tc, times=3)
    assert len(api.pod_names(NS)) == 2
    env = {e["name"]: e["value"] for e in
           api.get_pod(NS, "elastic-trainer-0")["spec"]["containers"][0]["env"]}
    assert env["WORLD_SIZE"] == "2"
    assert env["TRAININGJOB_RENDEZVOUS_EPOCH"] == "1"


def test_resize_out_of_bounds_rejected(cluster):
    api, tc = cluster
    api.create_job(NS, make_job(replicas=2))
    sync(tc)
    resize(api, 16)  # > maxReplicas
    sync(tc)
    assert any(e["reason"] == "ValidationFailed" for e in api.events)
    assert len(api.pod_names(NS)) == 2  # unchanged


def test_edl_never_ignores_world_changes(cluster):
    api, tc = cluster
    job = make_job(replicas=2)
    job["spec"]["replicaSpecs"]["trainer"]["edlPolicy"] = "Never"
    del job["spec"]["replicaSpecs"]["trainer"]["minReplicas"]
    del job["spec"]["replicaSpecs"]["trainer"]["maxReplicas"]
    api.create_job(NS, job)
    sync(tc)
    api.set_all_pods_phase(NS, "Running")
    sync(tc)
    resize(api, 3)
    sync(tc)
    # non-elastic: reference behavior is plain gap-fill (no epoch bump)
    assert len(api.pod_names(NS)) == 3
    assert EPOCH_ANNOTATION not in job_of(api).annotations
