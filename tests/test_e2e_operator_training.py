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
