"""Expectations gate + stray-pod adoption
(reference: controller.go:390-404, pod.go:134-150, 489-494)."""
import copy

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.expectations import (
    ControllerExpectations,
)
from trainingjob_operator_amd.kube import objects as ko
from trainingjob_operator_amd.kube.fake import FakeKubeApi

NS = "default"


def mk_job(name="exp", replicas=2):
    return {
        "apiVersion": C.API_VERSION,
        "kind": C.CRD_KIND,
        "metadata": {"name": name, "namespace": NS},
        "spec": {
            "replicaSpecs": {
                "trainer": {
                    "replicas": replicas,
                    "template": {"spec": {"containers": [
                        {"name": "aitj-main", "image": "x",
                         "ports": [{"name": "aitj-p", "containerPort": 23456}]}
                    ]}},
                },
            },
        },
    }


def test_expectations_unit():
    e = ControllerExpectations()
    assert e.satisfied("k")
    e.expect_creation("k", "p0")
    assert not e.satisfied("k")
    e.creation_observed("k", "p0")
    assert e.satisfied("k")
    e.expect_deletion("k", "p1")
    assert not e.satisfied("k")
    # the fresh list no longer contains p1 -> deletion happened
    e.observe_list("k", {"p2"})
    assert e.satisfied("k")
    # TTL expiry can never deadlock
    e.expect_creation("k", "px")
    assert e.satisfied("k", now=1e18)


def test_sync_skipped_while_operations_in_flight():
    api = FakeKubeApi()
    api.create_job(NS, mk_job())
    c = TrainingJobController(api)
    c.sync_once(f"{NS}/exp")
    pods = api.list_pods(NS)
    assert len(pods) == 2
    # a create the controller believes is still in flight (e.g. an
    # ambiguous timeout) suppresses reconcile: delete a pod out from
    # under it AND raise a pending expectation -> no recreate this sync
    api.delete_pod(NS, "exp-trainer-1")
    c.expectations.expect_creation(f"{NS}/exp", "pod/exp-trainer-ghost")
    c.sync_once(f"{NS}/exp")
    assert len(api.list_pods(NS)) == 1      # gap NOT refilled
    # once observed, the next sync reconciles normally
    c.expectations.creation_observed(f"{NS}/exp", "pod/exp-trainer-ghost")
    c.sync_once(f"{NS}/exp")
    assert len(api.list_pods(NS)) == 2


def test_fresh_list_settles_expectations():
    api = FakeKubeApi()
    api.create_job(NS, mk_job())
    c = TrainingJobController(api)
    c.sync_once(f"{NS}/exp")
    # the create calls raised + settled expectations via the API ack and
    # the next sync's list; nothing pending
    assert c.expectations.satisfied(f"{NS}/exp")
    # a pending creation whose pod IS in the list settles at sync time
    c.expectations.expect_creation(f"{NS}/exp", "pod/exp-trainer-0")
    c.sync_once(f"{NS}/exp")
    assert c.expectations.satisfied(f"{NS}/exp")


def test_stray_pod_with_matching_labels_is_adopted():
    api = FakeKubeApi()
    api.create_job(NS, mk_job())
    c = TrainingJobController(api)
    c.sync_once(f"{NS}/exp")
    # orphan one pod: strip its ownerReferences (simulates a lost ref)
    pod = api.get_pod(NS, "exp-trainer-0")
    api.patch_pod_metadata(NS, "exp-trainer-0", {"ownerReferences": []})
    pod = api.get_pod(NS, "exp-trainer-0")
    assert not ko.controller_ref(pod)
    c.sync_once(f"{NS}/exp")
    pod = api.get_pod(NS, "exp-trainer-0")
    ref = ko.controller_ref(pod)
    assert ref and ref["kind"] == C.CRD_KIND and ref["name"] == "exp"
    # and it was re-adopted, not recreated: still exactly 2 pods
    assert len(api.list_pods(NS)) == 2


def test_foreign_controller_pod_not_adopted():
    api = FakeKubeApi()
    api.create_job(NS, mk_job(replicas=1))
    c = TrainingJobController(api)
    c.sync_once(f"{NS}/exp")
    # a pod with matching labels but ANOTHER controllerRef: left alone
    stray = copy.deepcopy(api.get_pod(NS, "exp-trainer-0"))
    stray["metadata"]["name"] = "foreign-pod"
    stray["metadata"]["ownerReferences"] = [{
        "apiVersion": "apps/v1", "kind": "ReplicaSet", "name": "other",
        "uid": "other-uid", "controller": True,
    }]
    stray["metadata"].pop("resourceVersion", None)
    api.create_pod(NS, stray)
    c.sync_once(f"{NS}/exp")
    pod = api.get_pod(NS, "foreign-pod")
    ref = ko.controller_ref(pod)
    assert ref["kind"] == "ReplicaSet"     # untouched
