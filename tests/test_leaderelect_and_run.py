"""Leader election over Leases + the threaded controller run loop end-to-end
against the fake API (watch -> queue -> worker -> pods created)."""
import threading
import time

import pytest

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.types import AITrainingJob, Phase
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.leaderelect import LeaderElector
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.kube.fake import FakeKubeApi

NS = "default"


def test_leader_election_single_holder():
    api = FakeKubeApi()
    a = LeaderElector(api, "kube-system", "tj-operator", identity="a")
    b = LeaderElector(api, "kube-system", "tj-operator", identity="b")
    now = 1000.0
    assert a.try_acquire_or_renew(now)
    assert not b.try_acquire_or_renew(now + 1)
    assert a.try_acquire_or_renew(now + 5)  # renew
    # a stops renewing; lease expires; b takes over with a transition bump
    assert b.try_acquire_or_renew(now + 5 + 16)
    lease = api.get_lease("kube-system", "tj-operator")
    assert lease["spec"]["holderIdentity"] == "b"
    assert lease["spec"]["leaseTransitions"] == 1


@pytest.mark.timeout(120)
def test_run_loop_reconciles_via_watch():
    api = FakeKubeApi()
    opts = OperatorOptions(thread_num=2, resync_period=0.2, gc_period=30,
                           leader_elect=False)
    tc = TrainingJobController(api, opts)
    stop = threading.Event()
    t = threading.Thread(target=tc.run, args=(stop,), daemon=True)
    t.start()
    try:
        api.create_job(NS, {
            "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
            "metadata": {"name": "w", "namespace": NS},
            "spec": {"replicaSpecs": {"trainer": {
                "replicas": 2,
                "template": {"spec": {"containers": [{
                    "name": "aitj-main",
                    "ports": [{"name": "aitj-p", "containerPort": 5000}],
                }]}}}}},
        })
        deadline = time.monotonic() + 20
        while time.monotonic() < deadline:
            if len(api.pod_names(NS)) == 2:
                break
            time.sleep(0.05)
        assert api.pod_names(NS) == ["w-trainer-0", "w-trainer-1"]
        # CRD was self-registered at startup
        assert C.CRD_NAME in api.crds
        # drive to Running through watch events alone
        api.set_all_pods_phase(NS, "Running")
        deadline = time.monotonic() + 20
        while time.monotonic() < deadline:
            job = AITrainingJob.from_dict(api.get_job(NS, "w"))
            if job.status.phase == Phase.RUNNING:
                break
            time.sleep(0.05)
        assert job.status.phase == Phase.RUNNING
    finally:
        stop.set()
        t.join(timeout=10)


@pytest.mark.timeout(120)
def test_run_loop_restart_dance_async():
    """The full two-sync restart dance driven purely by watch events through
    the threaded workers (no manual sync calls)."""
    api = FakeKubeApi()
    opts = OperatorOptions(thread_num=2, resync_period=0.1, gc_period=30,
                           leader_elect=False)
    tc = TrainingJobController(api, opts)
    stop = threading.Event()
    t = threading.Thread(target=tc.run, args=(stop,), daemon=True)
    t.start()
    try:
        api.create_job(NS, {
            "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
            "metadata": {"name": "rd", "namespace": NS},
            "spec": {"restartingExitCode": "137",
                     "replicaSpecs": {"trainer": {
                         "replicas": 2, "restartPolicy": "ExitCode",
                         "restartScope": "All", "restartLimit": 3,
                         "template": {"spec": {"containers": [{
                             "name": "aitj-main",
                             "ports": [{"name": "aitj-p",
                                        "containerPort": 5000}],
                         }]}}}}},
        })

        def wait_for(pred, what, timeout=30):
            deadline = time.monotonic() + timeout
            while time.monotonic() < deadline:
                if pred():
                    return
                time.sleep(0.05)
            raise AssertionError(f"timed out waiting for {what}")

        wait_for(lambda: len(api.pod_names(NS)) == 2, "pods created")
        api.set_all_pods_phase(NS, "Running")
        wait_for(lambda: AITrainingJob.from_dict(
            api.get_job(NS, "rd")).status.phase == Phase.RUNNING, "Running")

        # retryable failure -> whole world restarts with bumped count
        api.set_pod_phase(NS, "rd-trainer-1", "Failed", exit_code=137)
        wait_for(lambda: len(api.pod_names(NS)) == 2 and all(
            api.get_pod(NS, n)["metadata"]["labels"]["RestartCount"] == "1"
            for n in api.pod_names(NS)), "pods recreated with RestartCount=1")
        api.set_all_pods_phase(NS, "Running")
        wait_for(lambda: AITrainingJob.from_dict(
            api.get_job(NS, "rd")).status.phase == Phase.RUNNING,
            "Running again")
        job = AITrainingJob.from_dict(api.get_job(NS, "rd"))
        assert job.status.restart_counts["trainer"] == 1
    finally:
        stop.set()
        t.join(timeout=10)
