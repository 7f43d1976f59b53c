"""CLI arg-parsing smoke tests + multi-role env contract details."""
import subprocess
import sys

from trainingjob_operator_amd.api import constants as C
from trainingjob_operator_amd.api.defaults import set_defaults
from trainingjob_operator_amd.api.types import AITrainingJob
from trainingjob_operator_amd.controller.envinject import render_env


def _help(module):
    out = subprocess.run([sys.executable, "-m", module, "--help"],
                         capture_output=True, text=True, timeout=60)
    assert out.returncode == 0, out.stderr
    return out.stdout


def test_operator_server_help():
    text = _help("trainingjob_operator_amd.controller.server")
    for flag in ("--namespace", "--resync-period", "--thread-num",
                 "--creating-restart-period", "--enable-creating-failed",
                 "--metrics-port"):
        assert flag in text


def test_launcher_help():
    text = _help("trainingjob_operator_amd.launcher.main")
    for flag in ("--model", "--steps", "--ckpt-dir", "--seq-len"):
        assert flag in text


def test_node_agent_help():
    text = _help("trainingjob_operator_amd.agent.node_agent")
    assert "--node-name" in text


def test_multirole_env_contract():
    """PS/worker jobs: every pod sees BOTH roles' host lists (reference
    pod.go:553-598 iterates all replicaSpecs) but a role-LOCAL torch world."""
    job = set_defaults(AITrainingJob.from_dict({
        "metadata": {"name": "ps", "namespace": "ml"},
        "spec": {"replicaSpecs": {
            "pserver": {"replicas": 2, "template": {"spec": {"containers": [
                {"name": "aitj-ps",
                 "ports": [{"name": "aitj-grpc", "containerPort": 2222}]}]}}},
            "worker": {"replicas": 3, "template": {"spec": {"containers": [
                {"name": "aitj-w",
                 "ports": [{"name": "aitj-grpc", "containerPort": 2223}]}]}}},
        }},
    }))
    env = {e["name"]: e["value"]
           for e in render_env(job, "worker", index=1, restart_count=0)}
    # cross-role visibility (reference contract)
    assert env["PSERVER_INSTANCES"] == "ps-pserver-0.ml,ps-pserver-1.ml"
    assert env["PSERVER_HOSTS"] == "ps-pserver-0.ml:2222,ps-pserver-1.ml:2222"
    assert env["WORKER_INSTANCES_NUM"] == "3"
    assert env["WORKER_PORTS"] == "2223"
    # role-local torch world (MI355X extension)
    assert env["WORLD_SIZE"] == "3"          # worker role size, not 5
    assert env["RANK"] == "1"
    assert env["MASTER_ADDR"] == "ps-worker-0.ml"
    assert env["MASTER_PORT"] == "2223"      # the role's own aitj port


def test_operator_yaml_config(tmp_path):
    import argparse
    import os
    from trainingjob_operator_amd.controller.options import OperatorOptions
    cfgfile = os.path.join(str(tmp_path), "op.yaml")
    with open(cfgfile, "w") as f:
        f.write("namespace: prod\nresync_period: 30.0\n"
                "elastic_scaleup_interval: 120.0\n")
    ap = argparse.ArgumentParser()
    OperatorOptions.add_flags(ap)
    # file values land; explicit flags override
    o = OperatorOptions.from_args(ap.parse_args(["--config", cfgfile]))
    assert o.namespace == "prod" and o.resync_period == 30.0
    assert o.elastic_scaleup_interval == 120.0
    o2 = OperatorOptions.from_args(ap.parse_args(
        ["--config", cfgfile, "--namespace", "dev"]))
    assert o2.namespace == "dev" and o2.resync_period == 30.0
    # unknown keys rejected
    with open(cfgfile, "w") as f:
        f.write("not_an_option: 1\n")
    import pytest
    with pytest.raises(ValueError):
        OperatorOptions.from_yaml(cfgfile)


def test_launcher_mode_flag_validation(tmp_path, monkeypatch):
    """Invalid mode combinations fail fast with clear messages."""
    import os
    import subprocess
    import sys
    env = {k: v for k, v in os.environ.items()
           if k not in ("RANK", "WORLD_SIZE")}
    env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29741"})

    def run(*flags):
        return subprocess.run(
            [sys.executable, "-m",
             "trainingjob_operator_amd.launcher.main",
             "--model", "llama-tiny", "--steps", "1", "--seq-len", "32",
             "--ckpt-dir", str(tmp_path)] + list(flags),
            env=env, capture_output=True, text=True, timeout=300)

    r = run("--sp")                       # sp without tp
    assert r.returncode != 0 and "--sp requires --tp" in r.stderr
    r = run("--zero1", "--tp", "2")       # zero1 with tp
    assert r.returncode != 0 and "pure-DP" in r.stderr
    r = run("--vocab-parallel")           # vp without tp
    assert r.returncode != 0 and "pure-TP" in r.stderr
    r = run("--pp", "2", "--ep", "2")     # pp + ep flag clash
    assert r.returncode != 0 and "drop --ep" in r.stderr


def test_multi_gpu_pod_env_contract():
    """amd.com/gpu > 1: node-level rank contract (VERDICT round-1 #4)."""
    from trainingjob_operator_amd.api.types import AITrainingJob
    from trainingjob_operator_amd.controller.envinject import (
        gpus_per_pod, render_env,
    )
    job = AITrainingJob.from_dict({
        "apiVersion": "elasticdeeplearning.ai/v1", "kind": "AITrainingJob",
        "metadata": {"name": "mg", "namespace": "default"},
        "spec": {"replicaSpecs": {"trainer": {
            "replicas": 2,
            "template": {"spec": {"containers": [{
                "name": "aitj-main",
                "resources": {"limits": {"amd.com/gpu": "4"}},
                "ports": [{"name": "aitj-p", "containerPort": 23456}],
            }]}},
        }}},
    })
    spec = job.spec.replica_specs["trainer"]
    assert gpus_per_pod(spec) == 4
    env = {e["name"]: e["value"] for e in render_env(job, "trainer", 1, 0)}
    assert env["WORLD_SIZE"] == "8"          # 2 pods x 4 GPUs
    assert env["RANK"] == "4"                # base rank of pod index 1
    assert env["NODE_RANK"] == "1"
    assert env["NPROC_PER_NODE"] == "4"
    assert env["LOCAL_WORLD_SIZE"] == "4"
    assert "LOCAL_RANK" not in env           # torchrun assigns per-process


def test_single_gpu_pod_env_contract_unchanged():
    from trainingjob_operator_amd.api.types import AITrainingJob
    from trainingjob_operator_amd.controller.envinject import render_env
    job = AITrainingJob.from_dict({
        "apiVersion": "elasticdeeplearning.ai/v1", "kind": "AITrainingJob",
        "metadata": {"name": "sg", "namespace": "default"},
        "spec": {"replicaSpecs": {"trainer": {
            "replicas": 3,
            "template": {"spec": {"containers": [{
                "name": "aitj-main",
                "resources": {"limits": {"amd.com/gpu": "1"}},
                "ports": [{"name": "aitj-p", "containerPort": 23456}],
            }]}},
        }}},
    })
    env = {e["name"]: e["value"] for e in render_env(job, "trainer", 2, 0)}
    assert env["WORLD_SIZE"] == "3"
    assert env["RANK"] == "2"
    assert env["LOCAL_RANK"] == "0"
    assert env["NPROC_PER_NODE"] == "1"
