"""API schema round-trip / quirk-fidelity tests (reference types.go/replica.go)."""
import json

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.defaults import set_defaults
from trainingjob_operator_amd.api.types import (
    AITrainingJob,
    CleanPodPolicy,
    EndingPolicy,
    Phase,
    PHASE_REASON,
    RestartPolicy,
    RestartScope,
    gen_general_name,
)
from trainingjob_operator_amd.api.validation import validate, validate_or_raise, ValidationError

EXAMPLE = {
    "apiVersion": "elasticdeeplearning.ai/v1",
    "kind": "AITrainingJob",
    "metadata": {"name": "paddle-mnist", "namespace": "default"},
    "spec": {
        "restartingExitCode": "137,128",
        "cleanPodPolicy": "None",
        "failPolicy": "Any",
        "completePolicy": "All",
        "replicaSpecs": {
            "trainer": {
                "replicas": 2,
                "restartLimit": 3,
                "restartPolicy": "ExitCode",
                "restartScope": "All",
                "template": {
                    "spec": {
                        "containers": [
                            {
                                "name": "aitj-trainer",
                                "image": "paddle:latest",
                                "ports": [{"name": "aitj-port", "containerPort": 8888}],
                            }
                        ]
                    }
                },
            }
        },
    },
}


def test_crd_identity():
    assert C.CRD_GROUP == "elasticdeeplearning.ai"
    assert C.CRD_KIND == "AITrainingJob"
    assert C.CRD_PLURAL == "aitrainingjobs"
    assert C.CRD_SHORT_NAME == "aitj"
    assert C.API_VERSION == "elasticdeeplearning.ai/v1"
    assert C.CRD_NAME == "aitrainingjobs.elasticdeeplearning.ai"


def test_phase_strings_match_reference():
    # types.go:100-124 — note "Succeed", not "Succeeded"
    assert Phase.SUCCEEDED == "Succeed"
    assert Phase.NODE_FAIL == "NodeFail"
    assert PHASE_REASON[Phase.SUCCEEDED] == "TrainingJobSucceed"
    assert PHASE_REASON[Phase.NONE] == ""


def test_env_names_match_reference():
    assert C.ENV_REPLICA_RESTART_COUNT == "TRAININGJOB_REPLICA_RESTARTCOUNT"
    assert C.ENV_SERVICE == "TRAININGJOB_SERVICE"
    assert C.ENV_PORTS == "TRAININGJOB_PORTS"
    assert C.CONTAINER_PREFIX == "aitj-"


def test_roundtrip_example():
    job = AITrainingJob.from_dict(EXAMPLE)
    assert job.name == "paddle-mnist"
    assert job.spec.restarting_exit_code == "137,128"
    assert job.spec.replica_specs["trainer"].replicas == 2
    d = job.to_dict()
    assert d["spec"]["restartingExitCode"] == "137,128"
    assert d["spec"]["replicaSpecs"]["trainer"]["restartPolicy"] == "ExitCode"
    # round-trip stability
    assert AITrainingJob.from_dict(d).to_dict() == d
    json.dumps(d)  # serializable


def test_status_json_quirks():
    job = AITrainingJob.from_dict(EXAMPLE)
    job.status.restart_counts = {"trainer": 2}
    job.status.restart_replica_name = "trainer"
    d = job.to_dict()
    # Go tag json:"RestartCount,,omitempty" -> "RestartCount" (types.go:84)
    assert d["status"]["RestartCount"] == {"trainer": 2}
    # untagged Go field -> exact field name, never omitted (types.go:86)
    assert d["status"]["RestartReplicaName"] == "trainer"
    job.status.restart_replica_name = ""
    job.status.restart_counts = {}
    d = job.to_dict()
    assert d["status"]["RestartReplicaName"] == ""
    assert "RestartCount" not in d["status"]  # omitempty


def test_defaults():
    job = AITrainingJob.from_dict({
        "metadata": {"name": "j"},
        "spec": {"replicaSpecs": {"trainer": {"template": {"spec": {"containers": [
            {"name": "aitj-x"}]}}}}},
    })
    set_defaults(job)
    assert job.spec.clean_pod_policy == CleanPodPolicy.ALL
    assert job.spec.fail_policy == EndingPolicy.ANY
    assert job.spec.complete_policy == EndingPolicy.ALL
    rs = job.spec.replica_specs["trainer"]
    assert rs.replicas == 1
    assert rs.restart_policy == RestartPolicy.NEVER
    assert rs.restart_scope == RestartScope.ALL
    assert rs.fail_policy == EndingPolicy.ANY
    assert rs.complete_policy == EndingPolicy.ALL


def test_defaults_keep_user_values():
    job = AITrainingJob.from_dict(EXAMPLE)
    set_defaults(job)
    assert job.spec.clean_pod_policy == "None"
    assert job.spec.replica_specs["trainer"].restart_policy == "ExitCode"


def test_elastic_defaults():
    job = AITrainingJob.from_dict({
        "metadata": {"name": "j"},
        "spec": {"replicaSpecs": {"trainer": {
            "replicas": 4, "edlPolicy": "Auto",
            "template": {"spec": {"containers": [{"name": "aitj-x"}]}}}}},
    })
    set_defaults(job)
    rs = job.spec.replica_specs["trainer"]
    assert rs.min_replicas == 4 and rs.max_replicas == 4


def test_validation_ok():
    job = set_defaults(AITrainingJob.from_dict(EXAMPLE))
    assert validate(job) == []


def test_validation_catches_errors():
    bad = AITrainingJob.from_dict({
        "metadata": {"name": "Bad_Name"},
        "spec": {
            "restartingExitCode": "137,x",
            "timeLimit": -5,
            "failPolicy": "Sometimes",
            "replicaSpecs": {
                "trainer": {
                    "replicas": 8, "minReplicas": 4, "maxReplicas": 6,
                    "restartPolicy": "Whenever",
                    "template": {"spec": {"containers": [{"name": "main"}]}},
                }
            },
        },
    })
    errs = validate(bad)
    joined = "\n".join(errs)
    assert "DNS-1123" in joined
    assert "restartingExitCode" in joined
    assert "timeLimit" in joined
    assert "failPolicy" in joined
    assert "restartPolicy" in joined
    assert "maxReplicas" in joined
    assert "aitj-" in joined
    try:
        validate_or_raise(bad)
        assert False, "should raise"
    except ValidationError as e:
        assert len(e.errors) >= 6


def test_gen_general_name():
    # reference: trainingjob.go:12-15
    assert gen_general_name("job", "trainer", 0) == "job-trainer-0"


def test_k8stime_roundtrip():
    from trainingjob_operator_amd.utils.k8stime import format_time, parse_time
    t = 1700000000.0
    assert parse_time(format_time(t)) == t
    assert parse_time("2023-11-14T22:13:20Z") == t
    assert parse_time("2023-11-14T22:13:20.500Z") == t + 0.5
    assert parse_time("2023-11-14T23:13:20+01:00") == t
    assert parse_time("2023-11-14T21:13:20-01:00") == t
    assert parse_time(None) is None
    assert parse_time("") is None
    assert parse_time(12.5) == 12.5


def test_example_manifests_validate():
    """Every shipped example AITrainingJob must parse, default, and pass
    the validator (keeps manifests/ honest as validation evolves)."""
    import glob
    import os

    import yaml

    from trainingjob_operator_amd.api.defaults import set_defaults
    from trainingjob_operator_amd.api.types import AITrainingJob
    from trainingjob_operator_amd.api.validation import validate
    root = os.path.join(os.path.dirname(__file__), "..", "manifests",
                        "examples")
    found = 0
    for f in sorted(glob.glob(os.path.join(root, "*.yaml"))):
        with open(f) as fh:
            d = yaml.safe_load(fh)
        if d.get("kind") != "AITrainingJob":
            continue
        job = AITrainingJob.from_dict(d)
        set_defaults(job)
        assert validate(job) == [], f
        found += 1
    assert found >= 8
