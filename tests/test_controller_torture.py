"""Randomized controller torture test: a seeded stream of cluster events
(pod failures with random exit codes, node health flips, elastic resizes,
completions, deletions) driven against the fake API with invariants checked
after every sync. The reference has nothing like this (zero tests)."""
import random

import pytest

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.types import (
    AITrainingJob, ENDING_PHASES, Phase,
)
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.kube.fake import FakeKubeApi

NS = "default"
ALL_PHASES = {Phase.NONE, Phase.PENDING, Phase.CREATING, Phase.RUNNING,
              Phase.SUCCEEDED, Phase.FAILED, Phase.TIMEOUT,
              Phase.RESTARTING, Phase.TERMINATING, Phase.PREEMPTED,
              Phase.NODE_FAIL}


def make_job(rng):
    return {
        "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
        "metadata": {"name": "torture", "namespace": NS},
        "spec": {
            "restartingExitCode": "137,128",
            "replicaSpecs": {"trainer": {
                "replicas": rng.randint(1, 4),
                "minReplicas": 1,
                "maxReplicas": 8,
                "edlPolicy": "Manual",
                "restartLimit": rng.randint(1, 5),
                "restartPolicy": rng.choice(
                    ["Always", "OnFailure", "ExitCode", "OnNodeFail",
                     "OnNodeFailWithExitCode"]),
                "restartScope": rng.choice(["All", "Replica", "Pod"]),
                "template": {"spec": {"containers": [{
                    "name": "aitj-main",
                    "ports": [{"name": "aitj-p", "containerPort": 5000}],
                }]}},
            }},
        },
    }


def check_invariants(api: FakeKubeApi, history):
    job = AITrainingJob.from_dict(api.get_job(NS, "torture"))
    # 1. phase is always a legal phase
    assert job.status.phase in ALL_PHASES, job.status.phase
    # 2. restart counts never decrease
    rc = job.status.restart_counts.get("trainer", 0)
    assert rc >= history.get("rc", 0), "restart count decreased"
    history["rc"] = rc
    # 3. terminal phases are sticky
    if history.get("terminal"):
        assert job.status.phase == history["terminal"], \
            f"left terminal phase {history['terminal']} -> {job.status.phase}"
    if job.status.phase in ENDING_PHASES:
        history["terminal"] = job.status.phase
    # 4. pods never exceed the declared replica count
    replicas = job.spec.replica_specs["trainer"].replicas or 0
    assert len(api.pod_names(NS)) <= max(replicas, history.get("prev_replicas", replicas)), \
        "more pods than replicas"
    history["prev_replicas"] = replicas
    # 5. every pod owned, labeled, and env-injected
    for name in api.pod_names(NS):
        pod = api.get_pod(NS, name)
        labels = pod["metadata"]["labels"]
        assert labels[C.LABEL_GROUP_NAME] == C.CRD_GROUP
        env = {e["name"] for e in pod["spec"]["containers"][0]["env"]}
        assert "RANK" in env and "WORLD_SIZE" in env
    return job


@pytest.mark.parametrize("seed", [1, 7, 42, 77, 123, 1234, 4242, 9999])
def test_torture(seed):
    rng = random.Random(seed)
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    api.create_job(NS, make_job(rng))
    history = {}

    for step in range(120):
        action = rng.random()
        pods = api.pod_names(NS)
        try:
            if action < 0.35 and pods:
                # drive pod lifecycle forward
                name = rng.choice(pods)
                api.set_pod_phase(NS, name, rng.choice(
                    ["Running", "Running", "Running", "Succeeded"]))
            elif action < 0.50 and pods:
                # fault: random exit code (retryable or not)
                name = rng.choice(pods)
                api.set_pod_phase(NS, name, "Failed",
                                  exit_code=rng.choice([137, 128, 1, 0]))
            elif action < 0.58:
                api.set_node_ready("node-0", rng.random() < 0.8)
            elif action < 0.70:
                # elastic resize within [min, max]
                j = api.get_job(NS, "torture")
                j["spec"]["replicaSpecs"]["trainer"]["replicas"] = \
                    rng.randint(1, 8)
                api.update_job(NS, "torture", j)
            elif action < 0.75 and pods:
                api.delete_pod(NS, rng.choice(pods))
        except KeyError:
            pass  # pod vanished between list and action — fine

        tc.sync_once(f"{NS}/torture")
        job = check_invariants(api, history)
        if job.status.phase in ENDING_PHASES and \
                job.spec.clean_pod_policy != "None" and \
                not api.pod_names(NS):
            break

    # drain: let the controller settle with no more faults
    api.set_node_ready("node-0", True)
    for _ in range(10):
        tc.sync_once(f"{NS}/torture")
        check_invariants(api, history)


def make_multirole_job(rng):
    return {
        "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
        "metadata": {"name": "torture", "namespace": NS},
        "spec": {
            "restartingExitCode": "137,128",
            "completePolicy": "Any",
            "replicaSpecs": {
                "pserver": {
                    "replicas": rng.randint(1, 2),
                    "restartPolicy": "OnFailure",
                    "restartScope": "Replica",
                    "restartLimit": 3,
                    "completePolicy": "None",
                    "template": {"spec": {"containers": [{
                        "name": "aitj-ps",
                        "ports": [{"name": "aitj-p",
                                   "containerPort": 6000}]}]}},
                },
                "worker": {
                    "replicas": rng.randint(1, 3),
                    "restartPolicy": rng.choice(["ExitCode", "Always"]),
                    "restartScope": rng.choice(["All", "Pod"]),
                    "restartLimit": 3,
                    "template": {"spec": {"containers": [{
                        "name": "aitj-w",
                        "ports": [{"name": "aitj-p",
                                   "containerPort": 5000}]}]}},
                },
            },
        },
    }


@pytest.mark.parametrize("seed", [5, 55, 555])
def test_torture_multirole(seed):
    rng = random.Random(seed)
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    api.create_job(NS, make_multirole_job(rng))
    history = {}

    for step in range(100):
        action = rng.random()
        pods = api.pod_names(NS)
        try:
            if action < 0.4 and pods:
                api.set_pod_phase(NS, rng.choice(pods), rng.choice(
                    ["Running", "Running", "Succeeded"]))
            elif action < 0.55 and pods:
                api.set_pod_phase(NS, rng.choice(pods), "Failed",
                                  exit_code=rng.choice([137, 1]))
            elif action < 0.6:
                api.set_node_ready("node-0", rng.random() < 0.85)
        except KeyError:
            pass
        tc.sync_once(f"{NS}/torture")
        job = AITrainingJob.from_dict(api.get_job(NS, "torture"))
        assert job.status.phase in ALL_PHASES
        # per-role restart counts monotonic
        for rt in ("pserver", "worker"):
            rc = job.status.restart_counts.get(rt, 0)
            assert rc >= history.get(rt, 0)
            history[rt] = rc
        if history.get("terminal"):
            assert job.status.phase == history["terminal"]
        from trainingjob_operator_amd.api.types import ENDING_PHASES as EP
        if job.status.phase in EP:
            history["terminal"] = job.status.phase
    api.set_node_ready("node-0", True)
    for _ in range(6):
        tc.sync_once(f"{NS}/torture")


def make_auto_job(rng):
    j = make_job(rng)
    rs = j["spec"]["replicaSpecs"]["trainer"]
    rs["edlPolicy"] = "Auto"
    rs["minReplicas"] = 1
    rs["maxReplicas"] = 6
    rs["replicas"] = rng.randint(1, 4)
    return j


@pytest.mark.parametrize("seed", [3, 17, 404, 8080])
def test_torture_auto_elastic(seed):
    """EdlPolicy=Auto under a randomized event stream with a simulated
    clock: unschedulable pods, faults, node flips. The controller's chosen
    target must always stay inside [min, max] and never materialize more
    pods than maxReplicas."""
    from trainingjob_operator_amd.controller.pods import TARGET_ANNOTATION
    rng = random.Random(seed)
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions(
        elastic_unschedulable_grace=5.0, elastic_scaleup_interval=20.0))
    api.create_job(NS, make_auto_job(rng))
    t = 1_700_000_000.0
    history = {}

    for step in range(120):
        t += rng.uniform(1.0, 10.0)
        action = rng.random()
        pods = api.pod_names(NS)
        try:
            if action < 0.35 and pods:
                api.set_pod_phase(NS, rng.choice(pods), rng.choice(
                    ["Running", "Running", "Running", "Succeeded"]))
            elif action < 0.50 and pods:
                api.set_pod_phase(NS, rng.choice(pods), "Failed",
                                  exit_code=rng.choice([137, 128, 1]))
            elif action < 0.62 and pods:
                # scheduler pressure: a pod cannot land anywhere
                api.set_pod_unschedulable(NS, rng.choice(pods),
                                          since=t - rng.uniform(0, 10))
            elif action < 0.70:
                api.set_node_ready("node-0", rng.random() < 0.8)
        except KeyError:
            pass

        tc.sync_once(f"{NS}/torture", now=t)
        job = AITrainingJob.from_dict(api.get_job(NS, "torture"))
        assert job.status.phase in ALL_PHASES
        rc = job.status.restart_counts.get("trainer", 0)
        assert rc >= history.get("rc", 0)
        history["rc"] = rc
        tgt = job.annotations.get(f"{TARGET_ANNOTATION}-trainer")
        if tgt is not None:
            assert 1 <= int(tgt) <= 6, f"target {tgt} out of [min,max]"
        assert len(api.pod_names(NS)) <= 6, "more pods than maxReplicas"
        if job.status.phase in ENDING_PHASES and not api.pod_names(NS):
            break

    # settle
    api.set_node_ready("node-0", True)
    for _ in range(10):
        t += 5.0
        tc.sync_once(f"{NS}/torture", now=t)
        assert len(api.pod_names(NS)) <= 6
