"""The full restart-policy x scope x exit-code matrix driven through the
CONTROLLER (not just the pure engine): for every combination, a pod failure
must produce exactly the reference's restart-or-fail outcome."""
import pytest

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.types import AITrainingJob, Phase
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.kube.fake import FakeKubeApi

NS = "default"

POLICIES = ["Always", "OnFailure", "OnNodeFail", "Never", "ExitCode",
            "OnNodeFailWithExitCode"]
SCOPES = ["All", "Replica", "Pod"]


def expected_restart_on_pod_failure(policy, code):
    """reference pod.go:385-405: pod-failure restarts under Always/OnFailure
    always, under *ExitCode policies only for retryable codes."""
    if policy in ("Always", "OnFailure"):
        return True
    if policy in ("ExitCode", "OnNodeFailWithExitCode"):
        return code in (137, 128)
    return False


def drive(policy, scope, code):
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    api.create_job(NS, {
        "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
        "metadata": {"name": "m", "namespace": NS},
        "spec": {"restartingExitCode": "137,128",
                 "replicaSpecs": {"trainer": {
                     "replicas": 2, "restartPolicy": policy,
                     "restartScope": scope, "restartLimit": 3,
                     "template": {"spec": {"containers": [{
                         "name": "aitj-main",
                         "ports": [{"name": "aitj-p",
                                    "containerPort": 5000}]}]}}}}},
    })
    tc.sync_once(f"{NS}/m")
    api.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/m")
    api.set_pod_phase(NS, "m-trainer-1", "Failed", exit_code=code)
    for _ in range(4):
        tc.sync_once(f"{NS}/m")
    return api, AITrainingJob.from_dict(api.get_job(NS, "m"))


@pytest.mark.parametrize("policy", POLICIES)
@pytest.mark.parametrize("scope", SCOPES)
@pytest.mark.parametrize("code", [137, 1])
def test_pod_failure_matrix(policy, scope, code):
    api, job = drive(policy, scope, code)
    if expected_restart_on_pod_failure(policy, code):
        # restarted: both pods back, count bumped, job not failed
        assert job.status.restart_counts["trainer"] == 1, \
            (policy, scope, code, job.status.restart_counts)
        assert len(api.pod_names(NS)) == 2
        assert job.status.phase not in (Phase.FAILED, Phase.SUCCEEDED)
        # All and Replica scope both restart the whole (single) role ->
        # every pod recreated at RestartCount 1; Pod scope recreates only
        # the failed one (survivor keeps its RestartCount 0 label,
        # reference pod.go:218-226)
        counts = sorted(
            api.get_pod(NS, n)["metadata"]["labels"]["RestartCount"]
            for n in api.pod_names(NS))
        if scope in ("All", "Replica"):
            assert counts == ["1", "1"]
        else:
            assert counts == ["0", "1"]
    else:
        # failPolicy Any (default): job fails
        assert job.status.phase == Phase.FAILED, (policy, scope, code)
        assert job.status.restart_counts.get("trainer", 0) == 0
