"""Garbage collector sweep + workqueue semantics."""
import time

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.controller.gc import GarbageCollector
from trainingjob_operator_amd.kube.fake import FakeKubeApi
from trainingjob_operator_amd.kube.workqueue import RateLimitedQueue
from trainingjob_operator_amd.utils.k8stime import format_time

NS = "default"


def orphan_pod(name, owner="gone-job", deletion_ts=None, labels=True):
    pod = {
        "metadata": {
            "name": name,
            "namespace": NS,
            "labels": {C.LABEL_GROUP_NAME: C.CRD_GROUP} if labels else {},
            "ownerReferences": [{
                "kind": C.CRD_KIND, "name": owner, "controller": True,
            }],
        },
        "spec": {"containers": [{"name": "aitj-x"}]},
    }
    if deletion_ts is not None:
        pod["metadata"]["deletionTimestamp"] = deletion_ts
    return pod


def test_gc_deletes_orphans_keeps_owned():
    api = FakeKubeApi()
    api.auto_schedule = False
    api.create_job(NS, {"metadata": {"name": "alive", "namespace": NS},
                        "spec": {"replicaSpecs": {}}})
    api.create_pod(NS, orphan_pod("owned", owner="alive"))
    api.create_pod(NS, orphan_pod("orphan", owner="gone-job"))
    api.create_pod(NS, orphan_pod("not-ours", labels=False))
    gc = GarbageCollector(api)
    deleted = gc.clean_garbage_pods(time.time())
    assert deleted == 1
    assert api.pod_names(NS) == ["not-ours", "owned"]


def test_gc_force_deletes_expired_terminating():
    api = FakeKubeApi()
    api.auto_schedule = False
    past = format_time(time.time() - 60)
    api.create_pod(NS, orphan_pod("stuck", owner="alive",
                                  deletion_ts=past))
    api.create_job(NS, {"metadata": {"name": "alive", "namespace": NS},
                        "spec": {"replicaSpecs": {}}})
    gc = GarbageCollector(api)
    assert gc.clean_garbage_pods(time.time()) == 1
    deletes = [a for a in api.actions if a[0] == "delete"]
    assert deletes[-1][4] == 0  # grace 0


def test_workqueue_dedup_and_dirty():
    q = RateLimitedQueue()
    q.add("a")
    q.add("a")
    assert len(q) == 1
    key = q.get(timeout=1)
    assert key == "a"
    q.add("a")  # while processing -> dirty
    assert len(q) == 0
    q.done("a")
    assert q.get(timeout=1) == "a"  # requeued after done
    q.done("a")


def test_workqueue_delayed_and_ratelimited():
    q = RateLimitedQueue(base_delay=0.01, max_delay=1.0)
    q.add_after("later", 0.05)
    assert q.get(timeout=0.01) is None
    assert q.get(timeout=1.0) == "later"
    q.done("later")
    t0 = time.monotonic()
    q.add_rate_limited("x")
    q.add_rate_limited("x")  # dedup in queue; failure count grows
    assert q.get(timeout=1.0) == "x"
    q.done("x")
    q.forget("x")
