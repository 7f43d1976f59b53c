"""RealKubeApi over a real loopback socket (kube_http_server fixture):
validates paths, label selectors, the status flow, leases, node patches and
watch streaming — then runs the WHOLE controller through the HTTP client."""
import threading
import time

import pytest

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.types import AITrainingJob, Phase
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.kube.client import ApiError, RealKubeApi
from trainingjob_operator_amd.kube.fake import FakeKubeApi

from kube_http_server import MockKubeServer

NS = "default"


@pytest.fixture
def cluster():
    fake = FakeKubeApi()
    server = MockKubeServer(fake).start()
    api = RealKubeApi(base_url=server.url)
    yield fake, api
    server.stop()


def job_manifest(name="httpjob"):
    return {
        "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
        "metadata": {"name": name, "namespace": NS},
        "spec": {"replicaSpecs": {"trainer": {
            "replicas": 2,
            "template": {"spec": {"containers": [{
                "name": "aitj-main",
                "ports": [{"name": "aitj-p", "containerPort": 5000}],
            }]}}}}},
    }


def test_pod_crud_and_selectors(cluster):
    fake, api = cluster
    api.create_pod(NS, {"metadata": {"name": "p1",
                                     "labels": {"app": "x"}},
                        "spec": {"containers": []}})
    api.create_pod(NS, {"metadata": {"name": "p2",
                                     "labels": {"app": "y"}},
                        "spec": {"containers": []}})
    assert len(api.list_pods(NS)) == 2
    only_x = api.list_pods(NS, selector={"app": "x"})
    assert [p["metadata"]["name"] for p in only_x] == ["p1"]
    assert api.get_pod(NS, "p2")["metadata"]["name"] == "p2"
    api.delete_pod(NS, "p1", grace_period=0)
    assert fake.pod_names(NS) == ["p2"]
    with pytest.raises(ApiError) as ei:
        api.get_pod(NS, "nope")
    assert ei.value.not_found


def test_job_status_roundtrip_and_conflict(cluster):
    fake, api = cluster
    fake.create_job(NS, job_manifest())
    job = api.get_job(NS, "httpjob")
    # the CRD declares a status subresource: a main-resource PUT silently
    # DROPS .status (real API-server semantics, emulated by the fake)...
    job["status"] = {"phase": "Running"}
    updated = api.update_job(NS, "httpjob", job)
    assert api.get_job(NS, "httpjob").get("status", {}).get("phase") \
        != "Running"
    # ...and only the /status endpoint persists it
    updated["status"] = {"phase": "Running"}
    api.update_job_status(NS, "httpjob", updated)
    assert api.get_job(NS, "httpjob")["status"]["phase"] == "Running"
    # stale resourceVersion conflicts
    with pytest.raises(ApiError) as ei:
        api.update_job(NS, "httpjob", job)
    assert ei.value.conflict


def test_lease_and_node_paths(cluster):
    fake, api = cluster
    api.create_lease("kube-system", {
        "metadata": {"name": "lk", "namespace": "kube-system"},
        "spec": {"holderIdentity": "me"}})
    assert api.get_lease("kube-system", "lk")["spec"]["holderIdentity"] == "me"
    api.patch_node_status("node-0", {"conditions": [
        {"type": "EDLGPUHealthy", "status": "False", "message": "ECC"}]})
    api.annotate_node("node-0", {"k": "v"})
    node = api.get_node("node-0")
    conds = {c["type"]: c for c in node["status"]["conditions"]}
    assert conds["EDLGPUHealthy"]["status"] == "False"
    assert node["metadata"]["annotations"]["k"] == "v"


def test_watch_streams_events(cluster):
    fake, api = cluster
    seen = []
    stop = threading.Event()

    def consume():
        for evt, obj in api.watch_pods(NS, stop):
            seen.append((evt, obj["metadata"]["name"]))
            if len(seen) >= 2:
                stop.set()
                return

    t = threading.Thread(target=consume, daemon=True)
    t.start()
    time.sleep(0.3)
    fake.create_pod(NS, {"metadata": {"name": "w1"}, "spec": {"containers": []}})
    fake.create_pod(NS, {"metadata": {"name": "w2"}, "spec": {"containers": []}})
    t.join(timeout=10)
    stop.set()
    assert ("ADDED", "w1") in seen and ("ADDED", "w2") in seen


@pytest.mark.timeout(120)
def test_full_controller_over_http(cluster):
    """The whole reconcile lifecycle with RealKubeApi as the transport."""
    fake, api = cluster
    tc = TrainingJobController(api, OperatorOptions())
    fake.create_job(NS, job_manifest("e2e"))
    tc.sync_once(f"{NS}/e2e")
    assert fake.pod_names(NS) == ["e2e-trainer-0", "e2e-trainer-1"]
    assert fake.service_names(NS) == ["e2e-trainer-0", "e2e-trainer-1"]
    fake.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/e2e")
    assert AITrainingJob.from_dict(
        fake.get_job(NS, "e2e")).status.phase == Phase.RUNNING
    fake.set_all_pods_phase(NS, "Succeeded")
    tc.sync_once(f"{NS}/e2e")
    tc.sync_once(f"{NS}/e2e")
    job = AITrainingJob.from_dict(fake.get_job(NS, "e2e"))
    assert job.status.phase == Phase.SUCCEEDED


def test_pod_log_over_http(cluster):
    fake, api = cluster
    fake.create_pod("d", {"metadata": {"name": "p1", "namespace": "d"},
                          "spec": {"containers": []}})
    fake.set_pod_log("d", "p1", "line1\nline2\nline3")
    assert api.read_pod_log("d", "p1") == "line1\nline2\nline3"
    assert api.read_pod_log("d", "p1", tail_lines=1) == "line3"


def test_aitjctl_over_http(cluster, capsys):
    """The ops CLI works against the real REST client end-to-end."""
    fake, api = cluster
    fake.create_job("d", job_manifest("cli-job"))
    from trainingjob_operator_amd import cli
    assert cli.main(["get", "-n", "d"], api=api) == 0
    out = capsys.readouterr().out
    assert "cli-job" in out
    assert cli.main(["describe", "cli-job", "-n", "d"], api=api) == 0
    assert "Name:      cli-job" in capsys.readouterr().out


def test_list_events_over_http(cluster):
    fake, api = cluster
    fake.create_job("d", job_manifest("evjob"))
    fake.events.append({
        "metadata": {"name": "evjob.1", "namespace": "d"},
        "involvedObject": {"name": "evjob"},
        "type": "Normal", "reason": "Test", "message": "hello"})
    evs = api.list_events("d", involved_name="evjob")
    assert len(evs) == 1 and evs[0]["reason"] == "Test"
    assert api.list_events("d", involved_name="other") == []


def test_watch_410_relist_recovery(cluster):
    """resourceVersion expiry: the server answers a watch with ERROR/410;
    the client yields a synthetic RELIST, drops its rv, reconnects, and
    keeps streaming subsequent events (client-go re-list semantics)."""
    import threading

    fake, api = cluster
    fake.create_job(NS, job_manifest())
    events = []
    stop = threading.Event()
    fake.expire_next_watch = True

    def consume():
        for etype, obj in api.watch_jobs(NS, stop):
            events.append((etype, obj.get("metadata", {}).get("name")))
            if len(events) >= 2:
                stop.set()
                return

    t = threading.Thread(target=consume, daemon=True)
    t.start()
    deadline = __import__("time").monotonic() + 10
    while not events and __import__("time").monotonic() < deadline:
        __import__("time").sleep(0.05)
    assert events and events[0][0] == "RELIST", events
    # post-expiry events still arrive on the reconnected stream
    j = api.get_job(NS, "httpjob")
    j["metadata"]["labels"] = {"poke": "1"}
    api.update_job(NS, "httpjob", j)
    deadline = __import__("time").monotonic() + 10
    while len(events) < 2 and __import__("time").monotonic() < deadline:
        __import__("time").sleep(0.05)
    stop.set()
    assert len(events) >= 2 and events[1][1] == "httpjob", events
