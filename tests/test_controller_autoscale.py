"""EdlPolicy=Auto: the controller picks the effective world size inside
[minReplicas, maxReplicas] itself — scale-down when pods sit unschedulable
past the grace period, +1 capacity probes toward maxReplicas afterwards.
(New semantics: the reference declared edlPolicy/min/max but never read
them — SURVEY.md §C15.)"""
import time

import pytest

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.types import AITrainingJob, Phase
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.controller.pods import (
    TARGET_ANNOTATION, WORLD_SIZE_ANNOTATION,
)
from trainingjob_operator_amd.kube.fake import FakeKubeApi

NS = "default"


def make_job(replicas=4, mn=2, mx=4):
    return {
        "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
        "metadata": {"name": "auto", "namespace": NS},
        "spec": {"replicaSpecs": {"trainer": {
            "replicas": replicas, "minReplicas": mn, "maxReplicas": mx,
            "edlPolicy": "Auto",
            "restartPolicy": "OnFailure", "restartScope": "All",
            "template": {"spec": {"containers": [{
                "name": "aitj-main",
                "ports": [{"name": "aitj-p", "containerPort": 5000}],
            }]}},
        }}},
    }


def job_of(api):
    return AITrainingJob.from_dict(api.get_job(NS, "auto"))


def world_annotations(api):
    return {name: api.get_pod(NS, name)["metadata"]["annotations"]
            [WORLD_SIZE_ANNOTATION] for name in api.pod_names(NS)}


def test_auto_scales_down_on_unschedulable():
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions(
        elastic_unschedulable_grace=60.0))
    api.create_job(NS, make_job(replicas=4, mn=2, mx=4))
    t0 = time.time()
    tc.sync_once(f"{NS}/auto", now=t0)
    assert len(api.pod_names(NS)) == 4
    api.set_all_pods_phase(NS, "Running")
    # one replica cannot schedule (e.g. its GPU node was cordoned)
    api.set_pod_unschedulable(NS, "auto-trainer-3",
                              since=t0 - 120)  # stuck past the grace
    tc.sync_once(f"{NS}/auto", now=t0 + 1)
    j = job_of(api)
    assert j.annotations[f"{TARGET_ANNOTATION}-trainer"] == "3"
    assert j.status.restart_replica_name == "trainer"
    assert api.pod_names(NS) == []          # world restart at new size
    tc.sync_once(f"{NS}/auto", now=t0 + 2)  # wait-gate -> Restarting
    assert job_of(api).status.phase == Phase.RESTARTING
    tc.sync_once(f"{NS}/auto", now=t0 + 3)  # recreate at effective size
    assert len(api.pod_names(NS)) == 3
    assert set(world_annotations(api).values()) == {"3"}
    env = {e["name"]: e["value"] for e in api.get_pod(
        NS, "auto-trainer-0")["spec"]["containers"][0]["env"]}
    assert env["WORLD_SIZE"] == "3"
    # spec itself is never mutated server-side
    assert api.get_job(NS, "auto")["spec"]["replicaSpecs"]["trainer"][
        "replicas"] == 4


def test_auto_scales_down_no_lower_than_min():
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions(
        elastic_unschedulable_grace=60.0))
    api.create_job(NS, make_job(replicas=3, mn=3, mx=4))
    t0 = time.time()
    tc.sync_once(f"{NS}/auto", now=t0)
    api.set_all_pods_phase(NS, "Running")
    api.set_pod_unschedulable(NS, "auto-trainer-2", since=t0 - 120)
    tc.sync_once(f"{NS}/auto", now=t0 + 1)
    # already at minReplicas: no resize, world left to (re)schedule
    j = job_of(api)
    assert j.annotations[f"{TARGET_ANNOTATION}-trainer"] == "3"
    assert j.status.restart_replica_name == ""
    assert len(api.pod_names(NS)) == 3


def test_auto_capacity_probe_scales_up_to_max():
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions(
        elastic_scaleup_interval=100.0))
    api.create_job(NS, make_job(replicas=2, mn=2, mx=3))
    t0 = time.time()
    tc.sync_once(f"{NS}/auto", now=t0)       # creates 2, starts the clock
    api.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/auto", now=t0 + 50)  # interval not elapsed
    assert len(api.pod_names(NS)) == 2
    assert job_of(api).annotations[f"{TARGET_ANNOTATION}-trainer"] == "2"

    tc.sync_once(f"{NS}/auto", now=t0 + 150)  # probe fires: 2 -> 3
    j = job_of(api)
    assert j.annotations[f"{TARGET_ANNOTATION}-trainer"] == "3"
    assert api.pod_names(NS) == []
    tc.sync_once(f"{NS}/auto", now=t0 + 151)
    tc.sync_once(f"{NS}/auto", now=t0 + 152)
    assert len(api.pod_names(NS)) == 3
    api.set_all_pods_phase(NS, "Running")
    # at maxReplicas: no further probes no matter how long we wait
    tc.sync_once(f"{NS}/auto", now=t0 + 1000)
    assert len(api.pod_names(NS)) == 3
    assert job_of(api).annotations[f"{TARGET_ANNOTATION}-trainer"] == "3"


def test_auto_probe_retreats_when_capacity_still_missing():
    """Up-probe 2->3, the new pod cannot schedule, the loop backs off to
    the 2 that fit — the full optimistic-probe cycle."""
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions(
        elastic_unschedulable_grace=60.0, elastic_scaleup_interval=100.0))
    api.create_job(NS, make_job(replicas=2, mn=2, mx=4))
    t0 = time.time()
    tc.sync_once(f"{NS}/auto", now=t0)
    api.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/auto", now=t0 + 150)  # probe: target 3, restart
    tc.sync_once(f"{NS}/auto", now=t0 + 151)
    tc.sync_once(f"{NS}/auto", now=t0 + 152)
    assert len(api.pod_names(NS)) == 3
    api.set_all_pods_phase(NS, "Running")
    api.set_pod_unschedulable(NS, "auto-trainer-2", since=t0 + 152)
    # before the grace elapses nothing happens
    tc.sync_once(f"{NS}/auto", now=t0 + 160)
    assert job_of(api).annotations[f"{TARGET_ANNOTATION}-trainer"] == "3"
    # after the grace the loop retreats to the 2 scheduled replicas
    tc.sync_once(f"{NS}/auto", now=t0 + 251)
    j = job_of(api)
    assert j.annotations[f"{TARGET_ANNOTATION}-trainer"] == "2"
    tc.sync_once(f"{NS}/auto", now=t0 + 252)
    tc.sync_once(f"{NS}/auto", now=t0 + 253)
    assert len(api.pod_names(NS)) == 2
    assert set(world_annotations(api).values()) == {"2"}


def test_manual_policy_ignores_auto_annotations():
    """EdlPolicy=Manual never auto-resizes even with a stale target
    annotation lying around."""
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    j = make_job(replicas=2, mn=2, mx=4)
    j["spec"]["replicaSpecs"]["trainer"]["edlPolicy"] = "Manual"
    j["metadata"]["annotations"] = {f"{TARGET_ANNOTATION}-trainer": "4"}
    api.create_job(NS, j)
    t0 = time.time()
    tc.sync_once(f"{NS}/auto", now=t0)
    api.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/auto", now=t0 + 1000)
    assert len(api.pod_names(NS)) == 2


def test_metrics_count_auto_resizes():
    from trainingjob_operator_amd.controller.metrics import OperatorMetrics
    m = OperatorMetrics(port=0)
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions(
        elastic_unschedulable_grace=60.0), metrics=m)
    api.create_job(NS, make_job(replicas=4, mn=2, mx=4))
    t0 = time.time()
    tc.sync_once(f"{NS}/auto", now=t0)
    api.set_all_pods_phase(NS, "Running")
    api.set_pod_unschedulable(NS, "auto-trainer-3", since=t0 - 120)
    tc.sync_once(f"{NS}/auto", now=t0 + 1)
    down = m.elastic_resizes_total.labels(direction="down")
    assert down._value.get() == 1
    # the restart the resize triggers is counted too
    assert m.restarts_total.labels(scope="All")._value.get() == 1


def test_auto_role_beside_fixed_role():
    """Multi-role job: the Auto trainer role scales down while the fixed
    pserver role is untouched."""
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions(
        elastic_unschedulable_grace=60.0))
    api.create_job(NS, {
        "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
        "metadata": {"name": "auto", "namespace": NS},
        "spec": {"replicaSpecs": {
            "pserver": {
                "replicas": 1, "restartPolicy": "OnFailure",
                "restartScope": "Replica",
                "template": {"spec": {"containers": [{
                    "name": "aitj-ps",
                    "ports": [{"name": "aitj-p",
                               "containerPort": 6000}]}]}},
            },
            "trainer": {
                "replicas": 3, "minReplicas": 2, "maxReplicas": 4,
                "edlPolicy": "Auto",
                "restartPolicy": "OnFailure", "restartScope": "Replica",
                "template": {"spec": {"containers": [{
                    "name": "aitj-main",
                    "ports": [{"name": "aitj-p",
                               "containerPort": 5000}]}]}},
            },
        }},
    })
    t0 = time.time()
    tc.sync_once(f"{NS}/auto", now=t0)
    assert len(api.pod_names(NS)) == 4
    api.set_all_pods_phase(NS, "Running")
    api.set_pod_unschedulable(NS, "auto-trainer-2", since=t0 - 120)
    tc.sync_once(f"{NS}/auto", now=t0 + 1)
    j = job_of(api)
    assert j.annotations[f"{TARGET_ANNOTATION}-trainer"] == "2"
    assert f"{TARGET_ANNOTATION}-pserver" not in j.annotations
    # restart scope Replica: only the trainer role restarts
    tc.sync_once(f"{NS}/auto", now=t0 + 2)
    tc.sync_once(f"{NS}/auto", now=t0 + 3)
    names = api.pod_names(NS)
    assert "auto-pserver-0" in names
    assert sum(1 for n in names if "trainer" in n) == 2


def test_manual_patch_resets_auto_target():
    """Patching spec.replicas on an Auto role resets the control loop's
    target: explicit user intent outranks the last automatic choice."""
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions(
        elastic_unschedulable_grace=60.0))
    api.create_job(NS, make_job(replicas=4, mn=2, mx=4))
    t0 = time.time()
    tc.sync_once(f"{NS}/auto", now=t0)
    api.set_all_pods_phase(NS, "Running")
    api.set_pod_unschedulable(NS, "auto-trainer-3", since=t0 - 120)
    tc.sync_once(f"{NS}/auto", now=t0 + 1)       # auto-down to 3
    assert job_of(api).annotations[
        f"{TARGET_ANNOTATION}-trainer"] == "3"
    tc.sync_once(f"{NS}/auto", now=t0 + 2)
    tc.sync_once(f"{NS}/auto", now=t0 + 3)
    assert len(api.pod_names(NS)) == 3

    # the user patches spec.replicas to 2 (e.g. via aitjctl resize)
    j = api.get_job(NS, "auto")
    j["spec"]["replicaSpecs"]["trainer"]["replicas"] = 2
    api.update_job(NS, "auto", j)
    tc.sync_once(f"{NS}/auto", now=t0 + 4)
    assert job_of(api).annotations[
        f"{TARGET_ANNOTATION}-trainer"] == "2"
    tc.sync_once(f"{NS}/auto", now=t0 + 5)
    tc.sync_once(f"{NS}/auto", now=t0 + 6)
    assert len(api.pod_names(NS)) == 2
