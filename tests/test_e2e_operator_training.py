"""End-to-end: operator-injected env -> real distributed training.

The controller creates the pods; we run the ACTUAL launcher as local
subprocesses with EXACTLY the env the controller injected (only the
rendezvous address is rewritten to loopback, standing in for the
headless-service DNS that needs a kubelet). Workers train over gloo,
exit 0, the fake kubelet reports Succeeded, and the job completes —
the full contract the reference delegates to opaque containers,
exercised for real (SURVEY.md §4 item 3's kind-cluster analog)."""
import os
import subprocess
import sys

import pytest

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.types import AITrainingJob, Phase
from trainingjob_operator_amd.controller.core import TrainingJobController
from trainingjob_operator_amd.controller.options import OperatorOptions
from trainingjob_operator_amd.kube.fake import FakeKubeApi

NS = "default"


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.mark.timeout(600)
def test_operator_env_bootstraps_real_training(tmp_path):
    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    api.create_job(NS, {
        "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
        "metadata": {"name": "e2e", "namespace": NS},
        "spec": {
            "completePolicy": "All",
            "replicaSpecs": {"trainer": {
                "replicas": 2, "restartPolicy": "OnFailure",
                "restartScope": "All",
                "template": {"spec": {"containers": [{
                    "name": "aitj-trainer",
                    "image": "trainingjob-operator-amd:latest",
                    "command": ["python", "-m",
                                "trainingjob_operator_amd.launcher.main"],
                    "ports": [{"name": "aitj-rccl",
                               "containerPort": 23456}],
                }]}},
            }},
        },
    })
    tc.sync_once(f"{NS}/e2e")
    pods = sorted(api.pod_names(NS))
    assert pods == ["e2e-trainer-0", "e2e-trainer-1"]

    # run the real launcher with the injected env (DNS -> loopback)
    port = _free_port()
    procs = []
    for name in pods:
        pod = api.get_pod(NS, name)
        env = {e["name"]: e.get("value", "")
               for e in pod["spec"]["containers"][0]["env"]}
        assert env["TRAININGJOB_NAME"] == "e2e"
        assert env["WORLD_SIZE"] == "2"
        full_env = {**os.environ, **env,
                    "MASTER_ADDR": "127.0.0.1",
                    "MASTER_PORT": str(port)}
        full_env.pop("CUDA_VISIBLE_DEVICES", None)
        procs.append((name, subprocess.Popen(
            [sys.executable, "-m",
             "trainingjob_operator_amd.launcher.main",
             "--model", "llama-tiny", "--steps", "3", "--seq-len", "32",
             "--grad-accum", "1", "--micro-batch", "1",
             "--ckpt-every", "2", "--log-every", "1",
             "--ckpt-dir", os.path.join(str(tmp_path), "ckpt")],
            env=full_env, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT, text=True)))

    for name, p in procs:
        out, _ = p.communicate(timeout=300)
        assert p.returncode == 0, f"{name}:\n{out[-2000:]}"

    # the kubelet-equivalent reports success; the controller completes
    api.set_all_pods_phase(NS, "Succeeded")
    tc.sync_once(f"{NS}/e2e")
    tc.sync_once(f"{NS}/e2e")
    job = AITrainingJob.from_dict(api.get_job(NS, "e2e"))
    assert job.status.phase == Phase.SUCCEEDED  # "Succeed"
    assert job.status.end_time is not None
    # training really happened: a checkpoint landed
    names = os.listdir(os.path.join(str(tmp_path), "ckpt"))
    assert any(n.startswith("ckpt_step") for n in names)


@pytest.mark.timeout(600)
def test_operator_elastic_resize_end_to_end(tmp_path):
    """BASELINE config 3 across the WHOLE stack: train at world 2 under
    the operator's env, patch spec.replicas to 1, let the controller run
    its epoch-bumped restart dance, and the recreated (real) worker
    resumes from checkpoint at the new world size and completes."""
    import time

    api = FakeKubeApi()
    tc = TrainingJobController(api, OperatorOptions())
    ckdir = os.path.join(str(tmp_path), "ckpt")
    api.create_job(NS, {
        "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
        "metadata": {"name": "el", "namespace": NS},
        "spec": {"completePolicy": "All", "replicaSpecs": {"trainer": {
            "replicas": 2, "minReplicas": 1, "maxReplicas": 4,
            "edlPolicy": "Manual", "restartPolicy": "OnFailure",
            "restartScope": "All", "restartLimit": 3,
            "template": {"spec": {"containers": [{
                "name": "aitj-trainer",
                "ports": [{"name": "aitj-rccl", "containerPort": 23456}],
            }]}}}}},
    })
    tc.sync_once(f"{NS}/el")

    def start_workers(port):
        procs = []
        for name in sorted(api.pod_names(NS)):
            pod = api.get_pod(NS, name)
            env = {e["name"]: e.get("value", "")
                   for e in pod["spec"]["containers"][0]["env"]}
            full_env = {**os.environ, **env,
                        "MASTER_ADDR": "127.0.0.1",
                        "MASTER_PORT": str(port)}
            procs.append(subprocess.Popen(
                [sys.executable, "-m",
                 "trainingjob_operator_amd.launcher.main",
                 "--model", "llama-tiny", "--steps", "6",
                 "--seq-len", "32", "--grad-accum", "1",
                 "--micro-batch", "1", "--ckpt-every", "1",
                 "--log-every", "1", "--ckpt-dir", ckdir],
                env=full_env, stdout=subprocess.PIPE,
                stderr=subprocess.STDOUT, text=True))
        return procs

    procs = start_workers(_free_port())
    # let the world make checkpointed progress
    deadline = time.monotonic() + 120
    while time.monotonic() < deadline:
        if os.path.isdir(ckdir) and any(
                n.startswith("ckpt_step") for n in os.listdir(ckdir)):
            break
        time.sleep(0.2)
    assert os.path.isdir(ckdir) and os.listdir(ckdir), "no checkpoint"
    api.set_all_pods_phase(NS, "Running")
    tc.sync_once(f"{NS}/el")

    # elastic resize 2 -> 1 (the operator kills the world; we stand in
    # for the kubelet's SIGTERM on the exact processes we own)
    j = api.get_job(NS, "el")
    j["spec"]["replicaSpecs"]["trainer"]["replicas"] = 1
    api.update_job(NS, "el", j)
    tc.sync_once(f"{NS}/el")            # stale world -> delete + mark
    assert api.pod_names(NS) == []
    for p in procs:
        p.terminate()
        p.wait(timeout=60)
    tc.sync_once(f"{NS}/el")            # wait gate -> Restarting
    tc.sync_once(f"{NS}/el")            # recreate at world 1
    assert api.pod_names(NS) == ["el-trainer-0"]
    pod = api.get_pod(NS, "el-trainer-0")
    env = {e["name"]: e.get("value", "")
           for e in pod["spec"]["containers"][0]["env"]}
    assert env["WORLD_SIZE"] == "1"
    assert env["TRAININGJOB_RENDEZVOUS_EPOCH"] == "1"

    (p,) = start_workers(_free_port())
    out, _ = p.communicate(timeout=300)
    assert p.returncode == 0, out[-2000:]
    assert "resumed from step" in out   # checkpoint carried across resize
    api.set_all_pods_phase(NS, "Succeeded")
    tc.sync_once(f"{NS}/el")
    tc.sync_once(f"{NS}/el")
    job = AITrainingJob.from_dict(api.get_job(NS, "el"))
    assert job.status.phase == Phase.SUCCEEDED


@pytest.mark.timeout(600)
def test_operator_path_restart_to_rejoin(tmp_path):
    """BASELINE config 4 through the CONTROLLER: SIGKILL a worker, the
    kubelet stand-in reports Failed(137), the operator runs the two-sync
    restart dance and recreates the world, and the rejoined workers make
    checkpointed progress. The measured wall time is the controller-path
    restart-to-rejoin metric (reported in BASELINE.md)."""
    from trainingjob_operator_amd.launcher.localrun import restart_benchmark
    times = restart_benchmark(model="llama-tiny", replicas=2, trials=1,
                              seq_len=32, ckpt_root=str(tmp_path))
    assert len(times) == 1 and times[0] > 0
    print(f"controller-path restart-to-rejoin: {times[0]:.2f}s")
