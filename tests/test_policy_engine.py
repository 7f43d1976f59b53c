"""Table-driven tests over the policy engine: the restart-policy x scope x
exit-code matrix plus condition and aggregation semantics the reference
implements but never tested."""
import pytest

from trainingjob_operator_amd.api.types import (
    AITrainingJob, Condition, EndingPolicy, Phase, RestartPolicy, RestartScope,
)
from trainingjob_operator_amd.policy.engine import (
    CreatingFailurePolicy,
    aggregate_job_ending,
    bump_restart_count,
    container_decision,
    count_replica_statuses,
    get_condition,
    is_failed_phase,
    is_job_completed,
    is_retryable_exit_code,
    restart_allowed,
    set_condition,
    update_job_conditions,
)

POL = CreatingFailurePolicy()
NOW = 1_700_000_000.0


def mkpod(phase="Running", node="node-a", container_state=None, name="p",
          start_time=None, reason="", message=""):
    cstatuses = []
    if container_state is not None:
        cstatuses = [{"name": "aitj-main", "state": container_state}]
    return {
        "metadata": {"name": name},
        "spec": {"nodeName": node} if node else {"spec": {}},
        "status": {
            "phase": phase,
            "containerStatuses": cstatuses,
            "startTime": start_time,
            "reason": reason,
            "message": message,
        },
    }


READY = {"node-a": True}


# --- is_retryable_exit_code (controller.go:442-462) ---

@pytest.mark.parametrize("codes,allowed,expect", [
    ([137], "137,128", True),
    ([137, 128], "137,128", True),
    ([137, 1], "137,128", False),   # AND over all codes
    ([], "137,128", False),          # empty list -> False
    ([0], "137,128", False),
    ([128], "128", True),
    ([137], "", False),
])
def test_retryable_exit_code(codes, allowed, expect):
    assert is_retryable_exit_code(codes, allowed) == expect


# --- container_decision: failure paths (pod.go:385-405) ---

@pytest.mark.parametrize("policy,code,expect_restart", [
    (RestartPolicy.ALWAYS, 1, True),
    (RestartPolicy.ON_FAILURE, 1, True),
    (RestartPolicy.NEVER, 1, False),
    (RestartPolicy.ON_NODE_FAIL, 1, False),
    (RestartPolicy.EXIT_CODE, 137, True),
    (RestartPolicy.EXIT_CODE, 1, False),
    (RestartPolicy.ON_NODE_FAIL_WITH_EXIT_CODE, 137, True),
    (RestartPolicy.ON_NODE_FAIL_WITH_EXIT_CODE, 1, False),
])
def test_failed_pod_restart_matrix(policy, code, expect_restart):
    pod = mkpod(phase="Failed",
                container_state={"terminated": {"exitCode": code, "reason": "Error"}})
    d = container_decision(pod, policy, "137,128", READY, None, POL, NOW)
    assert d.phase == Phase.FAILED
    assert d.is_restart == expect_restart
    if code != 0:
        assert "exitcode" in d.message


def test_failed_pod_message_falls_back_to_pod_reason():
    pod = mkpod(phase="Failed", reason="Evicted", message="node pressure")
    d = container_decision(pod, RestartPolicy.NEVER, "", READY, None, POL, NOW)
    assert d.phase == Phase.FAILED
    assert d.message == "Evicted, node pressure"


# --- node-fail path (pod.go:407-419) ---

@pytest.mark.parametrize("policy,expect_restart", [
    (RestartPolicy.ALWAYS, True),
    (RestartPolicy.ON_NODE_FAIL, True),
    (RestartPolicy.ON_NODE_FAIL_WITH_EXIT_CODE, True),
    (RestartPolicy.ON_FAILURE, False),
    (RestartPolicy.NEVER, False),
    (RestartPolicy.EXIT_CODE, False),
])
def test_node_fail_matrix(policy, expect_restart):
    pod = mkpod(phase="Running", node="dead-node",
                container_state={"running": {}})
    d = container_decision(pod, policy, "", READY, None, POL, NOW)
    assert d.phase == Phase.NODE_FAIL
    assert d.is_restart == expect_restart
    assert "dead-node" in d.message


# --- creating / succeeded paths (pod.go:421-435) ---

def test_creating_pod():
    pod = mkpod(phase="Pending",
                container_state={"waiting": {"reason": "ContainerCreating"}})
    d = container_decision(pod, RestartPolicy.NEVER, "", READY, None, POL, NOW)
    assert d.phase == Phase.CREATING
    assert d.message == "creating containers"
    assert not d.is_restart


def test_creating_error_reason_reported():
    pod = mkpod(phase="Pending",
                container_state={"waiting": {"reason": "ImagePullBackOff"}})
    d = container_decision(pod, RestartPolicy.NEVER, "", READY, None, POL, NOW)
    assert d.phase == Phase.CREATING
    assert "ImagePullBackOff" in d.message


def test_creating_error_restart_after_duration():
    # inside the creating-restart window AND pod stuck longer than
    # creating_duration -> restart (pod.go:355-368)
    pol = CreatingFailurePolicy(creating_restart_seconds=3600,
                                creating_duration_seconds=60)
    pod = mkpod(phase="Pending",
                container_state={"waiting": {"reason": "ErrImagePull"}},
                start_time=NOW - 120)
    d = container_decision(pod, RestartPolicy.NEVER, "", READY,
                           creating_condition_transition=NOW - 10,
                           policy=pol, now=NOW)
    assert d.phase == Phase.CREATING
    assert d.is_restart


def test_creating_error_fails_job_after_window():
    # past the window with enable_creating_failed -> Failed (pod.go:369-378)
    pol = CreatingFailurePolicy(creating_restart_seconds=5,
                                creating_duration_seconds=60,
                                enable_creating_failed=True)
    pod = mkpod(phase="Pending",
                container_state={"waiting": {"reason": "ErrImagePull"}},
                start_time=NOW - 120)
    d = container_decision(pod, RestartPolicy.NEVER, "", READY,
                           creating_condition_transition=NOW - 100,
                           policy=pol, now=NOW)
    assert d.phase == Phase.FAILED
    assert "retrying" in d.message


def test_succeeded_pod():
    pod = mkpod(phase="Succeeded",
                container_state={"terminated": {"exitCode": 0}})
    d = container_decision(pod, RestartPolicy.NEVER, "", READY, None, POL, NOW)
    assert d.phase == Phase.SUCCEEDED


def test_running_pod_no_phase():
    pod = mkpod(phase="Running", container_state={"running": {}})
    d = container_decision(pod, RestartPolicy.NEVER, "", READY, None, POL, NOW)
    assert d.phase == Phase.NONE
    assert not d.is_restart


# --- restart accounting (status.go:322-330, pod.go:215-216) ---

def _job(scope=RestartScope.ALL, limit=None):
    return AITrainingJob.from_dict({
        "metadata": {"name": "j"},
        "spec": {"replicaSpecs": {
            "trainer": {"replicas": 2, "restartScope": scope,
                        **({"restartLimit": limit} if limit is not None else {})},
            "pserver": {"replicas": 1, "restartScope": scope},
        }},
    })


def test_bump_scope_all_bumps_every_type():
    job = _job(RestartScope.ALL)
    bump_restart_count(job, "trainer")
    assert job.status.restart_counts == {"trainer": 1, "pserver": 1}


def test_bump_scope_pod_bumps_only_type():
    job = _job(RestartScope.POD)
    bump_restart_count(job, "trainer")
    assert job.status.restart_counts == {"trainer": 1}


def test_restart_limit():
    job = _job(RestartScope.POD, limit=2)
    assert restart_allowed(job, "trainer")
    job.status.restart_counts["trainer"] = 2
    assert not restart_allowed(job, "trainer")
    job2 = _job(RestartScope.POD)  # nil limit == unlimited
    job2.status.restart_counts["trainer"] = 99
    assert restart_allowed(job2, "trainer")


# --- replica counters (status.go:332-359) ---

def test_count_replica_statuses():
    pods = [
        mkpod(phase="Pending", node=None),
        mkpod(phase="Pending", node="node-a"),
        mkpod(phase="Running"),
        mkpod(phase="Succeeded"),
        mkpod(phase="Failed"),
        mkpod(phase="Unknown"),
    ]
    rs = count_replica_statuses(0, pods)
    assert (rs.pending, rs.scheduled, rs.active, rs.succeeded, rs.failed) == (1, 1, 1, 1, 2)
    rs2 = count_replica_statuses(1, pods)  # restarted role: Pending -> Restarting
    assert rs2.restarting == 2 and rs2.pending == 0 and rs2.scheduled == 0


# --- conditions (status.go:13-99) ---

def test_set_condition_append_and_flip():
    job = AITrainingJob.from_dict({"metadata": {"name": "j"}, "spec": {"replicaSpecs": {}}})
    update_job_conditions(job, Phase.PENDING, "waiting", NOW)
    update_job_conditions(job, Phase.CREATING, "creating", NOW + 1)
    update_job_conditions(job, Phase.CREATING, "creating more", NOW + 2)
    conds = job.status.conditions
    assert [c.type for c in conds] == [Phase.PENDING, Phase.CREATING]
    assert conds[0].status == "False"
    assert conds[1].status == "True"
    assert conds[1].message == "creating more"  # merged, not appended
    assert job.status.phase == Phase.CREATING


def test_conditions_freeze_after_completion():
    job = AITrainingJob.from_dict({"metadata": {"name": "j"}, "spec": {"replicaSpecs": {}}})
    update_job_conditions(job, Phase.SUCCEEDED, "done", NOW)
    assert is_job_completed(job.status)
    update_job_conditions(job, Phase.RUNNING, "zombie", NOW + 1)
    assert job.status.phase == Phase.SUCCEEDED
    assert len(job.status.conditions) == 1


# --- job-level aggregation (status.go:144-174) ---

def _agg_job(complete, fail):
    return AITrainingJob.from_dict({
        "metadata": {"name": "j"},
        "spec": {"completePolicy": complete, "failPolicy": fail,
                 "replicaSpecs": {"a": {"replicas": 1}, "b": {"replicas": 1}}},
    })


def test_aggregate_complete_any():
    job = _agg_job(EndingPolicy.ANY, EndingPolicy.ANY)
    d = aggregate_job_ending(job, {"a": Phase.SUCCEEDED, "b": Phase.NONE}, "")
    assert d.terminate and d.phase == Phase.SUCCEEDED


def test_aggregate_complete_beats_fail():
    job = _agg_job(EndingPolicy.ANY, EndingPolicy.ANY)
    d = aggregate_job_ending(job, {"a": Phase.SUCCEEDED, "b": Phase.FAILED}, "boom")
    assert d.terminate and d.phase == Phase.SUCCEEDED


def test_aggregate_fail_any():
    job = _agg_job(EndingPolicy.ALL, EndingPolicy.ANY)
    d = aggregate_job_ending(job, {"a": Phase.NODE_FAIL, "b": Phase.NONE}, "node died")
    assert d.terminate and d.phase == Phase.NODE_FAIL and d.message == "node died"


def test_aggregate_all_policies_wait():
    job = _agg_job(EndingPolicy.ALL, EndingPolicy.ALL)
    d = aggregate_job_ending(job, {"a": Phase.SUCCEEDED, "b": Phase.NONE}, "")
    assert not d.terminate
    d = aggregate_job_ending(job, {"a": Phase.SUCCEEDED, "b": Phase.SUCCEEDED}, "")
    assert d.terminate and d.phase == Phase.SUCCEEDED
    d = aggregate_job_ending(job, {"a": Phase.FAILED, "b": Phase.FAILED}, "x")
    assert d.terminate and d.phase == Phase.FAILED


def test_is_failed_phase():
    assert is_failed_phase(Phase.FAILED)
    assert is_failed_phase(Phase.TIMEOUT)
    assert is_failed_phase(Phase.NODE_FAIL)
    assert is_failed_phase(Phase.PREEMPTED)
    assert not is_failed_phase(Phase.SUCCEEDED)
    assert not is_failed_phase(Phase.RUNNING)
