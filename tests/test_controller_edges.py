"""Edge interactions: CleanPodPolicy=None with restarts, TimeLimit during
restart, GC across namespaces."""
import time

import pytest

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.types import AITrainingJob, Phase
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.kube.fake import FakeKubeApi

NS = "default"


def make_job(name="e", **spec_over):
    spec = {
        "restartingExitCode": "137",
        "replicaSpecs": {"trainer": {
            "replicas": 2, "restartPolicy": "ExitCode",
            "restartScope": "All", "restartLimit": 3,
            "template": {"spec": {"containers": [{
                "name": "aitj-main",
                "ports": [{"name": "aitj-p", "containerPort": 5000}],
            }]}}}},
    }
    spec.update(spec_over)
    return {"apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
            "metadata": {"name": name, "namespace": NS}, "spec": spec}


def test_clean_pod_none_failure_keeps_pods():
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    api.create_job(NS, make_job(cleanPodPolicy="None"))
    tc.sync_once(f"{NS}/e")
    api.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/e")
    # non-retryable failure -> job fails immediately, pods KEPT
    api.set_pod_phase(NS, "e-trainer-0", "Failed", exit_code=1)
    tc.sync_once(f"{NS}/e")
    job = AITrainingJob.from_dict(api.get_job(NS, "e"))
    assert job.status.phase == Phase.FAILED
    assert len(api.pod_names(NS)) == 2
    assert job.status.end_time is not None


def test_timelimit_fires_after_restart_cycle():
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    api.create_job(NS, make_job(timeLimit=300))
    t0 = time.time()
    tc.sync_once(f"{NS}/e", now=t0)
    api.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/e", now=t0 + 1)
    # restart at t+10 (retryable)
    api.set_pod_phase(NS, "e-trainer-1", "Failed", exit_code=137)
    tc.sync_once(f"{NS}/e", now=t0 + 10)
    tc.sync_once(f"{NS}/e", now=t0 + 11)
    tc.sync_once(f"{NS}/e", now=t0 + 12)
    api.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/e", now=t0 + 13)
    job = AITrainingJob.from_dict(api.get_job(NS, "e"))
    assert job.status.phase == Phase.RUNNING
    # startRunningTime survives the restart -> TimeLimit measured from the
    # FIRST run start (reference semantics: status.go:189-198)
    tc.sync_once(f"{NS}/e", now=t0 + 400)
    job = AITrainingJob.from_dict(api.get_job(NS, "e"))
    assert Phase.TIMEOUT in job.annotations
    tc.sync_once(f"{NS}/e", now=t0 + 401)
    assert AITrainingJob.from_dict(
        api.get_job(NS, "e")).status.phase == Phase.TIMEOUT


def test_gc_spares_other_namespaces_pods():
    from trainingjob_operator_amd.controller.gc import GarbageCollector
    api = FakeKubeApi()
    api.auto_schedule = False
    api.create_pod("other", {
        "metadata": {"name": "stray", "namespace": "other",
                     "labels": {C.LABEL_GROUP_NAME: C.CRD_GROUP},
                     "ownerReferences": [{"kind": C.CRD_KIND,
                                          "name": "gone",
                                          "controller": True}]},
        "spec": {"containers": []}})
    # namespace-scoped GC must not touch the other namespace
    gc = GarbageCollector(api, namespace=NS)
    assert gc.clean_garbage_pods(time.time()) == 0
    assert api.pod_names("other") == ["stray"]
    # cluster-wide GC does collect it
    gc_all = GarbageCollector(api)
    assert gc_all.clean_garbage_pods(time.time()) == 1


def test_fault_tolerant_widens_node_fail_restart():
    """spec.faultTolerant (dead in the reference) makes node loss
    retryable under ANY restart policy except Never."""
    for ft, expect_restart in ((True, True), (False, False)):
        api = FakeKubeApi()
        tc = TrainingJobController(api, OperatorOptions())
        api.create_job(NS, make_job(faultTolerant=ft))  # OnFailure-like?
        j = api.get_job(NS, "e")
        j["spec"]["replicaSpecs"]["trainer"]["restartPolicy"] = "OnFailure"
        api.update_job(NS, "e", j)
        tc.sync_once(f"{NS}/e")
        api.set_all_pods_phase(NS, "Running")
        tc.sync_once(f"{NS}/e")
        api.set_node_ready("node-0", False)
        tc.sync_once(f"{NS}/e")
        job = AITrainingJob.from_dict(api.get_job(NS, "e"))
        if expect_restart:
            assert job.status.restart_replica_name == "trainer"
        else:
            assert job.status.restart_replica_name == ""
            assert Phase.NODE_FAIL in job.annotations
            tc.sync_once(f"{NS}/e")  # deferred finalization
            job = AITrainingJob.from_dict(api.get_job(NS, "e"))
            assert job.status.phase == Phase.NODE_FAIL


def test_never_policy_ignores_fault_tolerant():
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    job = make_job(faultTolerant=True)
    job["spec"]["replicaSpecs"]["trainer"]["restartPolicy"] = "Never"
    api.create_job(NS, job)
    tc.sync_once(f"{NS}/e")
    api.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/e")
    api.set_node_ready("node-0", False)
    tc.sync_once(f"{NS}/e")
    job = AITrainingJob.from_dict(api.get_job(NS, "e"))
    assert job.status.restart_replica_name == ""
    assert Phase.NODE_FAIL in job.annotations
    tc.sync_once(f"{NS}/e")
    assert AITrainingJob.from_dict(
        api.get_job(NS, "e")).status.phase == Phase.NODE_FAIL


def test_last_reconcile_time_stamped_on_persist():
    from trainingjob_operator_amd.utils.k8stime import parse_time
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    api.create_job(NS, make_job())
    tc.sync_once(f"{NS}/e")
    job = AITrainingJob.from_dict(api.get_job(NS, "e"))
    t = parse_time(job.status.last_reconcile_time)
    assert t is not None and t > 0
