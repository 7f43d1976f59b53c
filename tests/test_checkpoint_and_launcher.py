"""Checkpointer unit tests + launcher env-contract fallbacks."""
import os

import torch

from trainingjob_operator_amd.launcher.checkpoint import Checkpointer
from trainingjob_operator_amd.parallel import dist_ctx
from trainingjob_operator_amd.training import TrainConfig, Trainer


def make_trainer():
    cfg = TrainConfig(model="llama-tiny", micro_batch=1, grad_accum=1,
                      seq_len=16, lr=1e-3)
    return Trainer(cfg)


def test_checkpoint_roundtrip(tmp_path):
    t1 = make_trainer()
    for _ in range(3):
        t1.train_step()
    ckpt = Checkpointer(str(tmp_path))
    path = ckpt.save_async(t1, blocking=True)
    assert os.path.exists(path)

    t2 = make_trainer()
    assert not torch.equal(t2.store.flat_param, t1.store.flat_param)
    step = ckpt.load_latest(t2)
    assert step == 3
    assert torch.equal(t2.store.flat_param, t1.store.flat_param)
    assert torch.equal(t2.opt.m, t1.opt.m)
    assert t2.opt.step_count == 3
    # resumed trainer keeps training
    t2.train_step()
    assert t2.opt.step_count == 4


def test_checkpoint_prune_keeps_latest(tmp_path):
    t = make_trainer()
    ckpt = Checkpointer(str(tmp_path), keep=2)
    for _ in range(4):
        t.train_step()
        ckpt.save_async(t, blocking=True)
    ckpts = ckpt.list_checkpoints()
    assert len(ckpts) == 2
    assert ckpt.latest().endswith("00000004.pt")


def test_checkpoint_async_then_wait(tmp_path):
    t = make_trainer()
    ckpt = Checkpointer(str(tmp_path))
    path = ckpt.save_async(t, blocking=False)
    ckpt.wait()
    assert os.path.exists(path)
    assert not os.path.exists(path + ".tmp")


def test_load_latest_empty_dir(tmp_path):
    t = make_trainer()
    assert Checkpointer(str(tmp_path)).load_latest(t) is None


def test_dist_ctx_torchrun_env(monkeypatch):
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT", "TRAININGJOB_REPLICA_NAME",
              "TRAININGJOB_REPLICA_INDEX", "TRAINER_INSTANCES",
              "TRAINER_INSTANCES_NUM"):
        monkeypatch.delenv(k, raising=False)
    monkeypatch.setenv("RANK", "3")
    monkeypatch.setenv("WORLD_SIZE", "8")
    monkeypatch.setenv("LOCAL_RANK", "3")
    monkeypatch.setenv("MASTER_ADDR", "10.0.0.1")
    monkeypatch.setenv("MASTER_PORT", "29500")
    ctx = dist_ctx.from_env()
    assert (ctx.rank, ctx.world_size, ctx.local_rank) == (3, 8, 3)
    assert ctx.master_addr == "10.0.0.1" and ctx.master_port == 29500


def test_dist_ctx_reference_contract_fallback(monkeypatch):
    """The operator's reference env alone (no torch-style vars) must be
    enough to bootstrap."""
    for k in ("RANK", "WORLD_SIZE", "LOCAL_RANK", "MASTER_ADDR",
              "MASTER_PORT"):
        monkeypatch.delenv(k, raising=False)
    monkeypatch.setenv("TRAININGJOB_REPLICA_NAME", "trainer")
    monkeypatch.setenv("TRAININGJOB_REPLICA_INDEX", "2")
    monkeypatch.setenv("TRAINER_INSTANCES_NUM", "4")
    monkeypatch.setenv("TRAINER_INSTANCES",
                       "j-trainer-0.ns,j-trainer-1.ns,j-trainer-2.ns,"
                       "j-trainer-3.ns")
    ctx = dist_ctx.from_env()
    assert (ctx.rank, ctx.world_size) == (2, 4)
    assert ctx.master_addr == "j-trainer-0.ns"


def test_worker_metrics_endpoint(tmp_path):
    """Launcher with --metrics-port exposes tokens/s on /metrics."""
    import subprocess, sys, time, urllib.request, socket
    s = socket.socket(); s.bind(("127.0.0.1", 0))
    mport = s.getsockname()[1]; s.close()
    s = socket.socket(); s.bind(("127.0.0.1", 0))
    dport = s.getsockname()[1]; s.close()
    env = dict(os.environ, RANK="0", WORLD_SIZE="1",
               MASTER_ADDR="127.0.0.1", MASTER_PORT=str(dport))
    proc = subprocess.Popen(
        [sys.executable, "-m", "trainingjob_operator_amd.launcher.main",
         "--model", "llama-tiny", "--steps", "100000", "--seq-len", "16",
         "--grad-accum", "1", "--log-every", "1", "--ckpt-every", "1000000",
         "--ckpt-dir", str(tmp_path), "--metrics-port", str(mport)],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True)
    try:
        body = ""
        deadline = time.monotonic() + 90
        while time.monotonic() < deadline:
            try:
                body = urllib.request.urlopen(
                    f"http://127.0.0.1:{mport}/metrics", timeout=2).read() \
                    .decode()
                if "aitj_worker_tokens_per_sec" in body and \
                        "aitj_worker_step" in body:
                    break
            except OSError:
                pass
            if proc.poll() is not None:
                break
            time.sleep(0.3)
        assert "aitj_worker_tokens_per_sec" in body, proc.stdout.read()[-800:]
    finally:
        proc.kill()
        proc.wait(timeout=30)


def test_launcher_eval_loop(tmp_path):
    """--eval-every runs a held-out eval (distinct seed stream) and traces
    it (subprocess: the tracer is a process-lifetime singleton)."""
    import json
    import subprocess
    import sys
    trace_path = os.path.join(str(tmp_path), "trace.jsonl")
    env = {k: v for k, v in os.environ.items()
           if k not in ("RANK", "WORLD_SIZE")}
    env["AITJ_TRACE"] = trace_path
    r = subprocess.run(
        [sys.executable, "-m", "trainingjob_operator_amd.launcher.main",
         "--model", "llama-tiny", "--steps", "4", "--seq-len", "32",
         "--grad-accum", "1", "--micro-batch", "1",
         "--ckpt-every", "100", "--log-every", "1",
         "--eval-every", "2", "--eval-batches", "2",
         "--ckpt-dir", str(tmp_path)],
        env=env, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    events = [json.loads(l) for l in open(trace_path)]
    evals = [e for e in events if e["kind"] == "eval"]
    assert len(evals) == 2                  # steps 2 and 4
    assert all("eval_loss" in e and e["eval_loss"] > 0 for e in evals)


def test_bench_contract_two_ranks(tmp_path):
    """The driver's exact multi-rank invocation shape: torchrun-style env,
    2 gloo ranks, rank 0 prints ONE JSON line with the whole-job value."""
    import json
    import subprocess
    import sys
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29733", "--no-python", "--",
         sys.executable, os.path.join(root, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--seq-len", "32", "--micro-batch", "1", "--grad-accum", "1"],
        capture_output=True, text=True, timeout=420, cwd=root)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, r.stdout[-1500:]      # exactly one JSON line
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == 2 and rec["steps"] == 2
    assert rec["scaling"] == "weak" and rec["dtype"] == "bf16"
    assert rec["value"] > 0 and rec["config"]["parallelism"] == "dp2"


def test_sigusr1_checkpoints_on_demand(tmp_path):
    """SIGUSR1 forces a checkpoint at the next step boundary (planned
    maintenance hook), independent of --ckpt-every."""
    import signal
    import subprocess
    import sys
    import time
    env = {k: v for k, v in os.environ.items()
           if k not in ("RANK", "WORLD_SIZE")}
    env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29747"})
    p = subprocess.Popen(
        [sys.executable, "-m", "trainingjob_operator_amd.launcher.main",
         "--model", "llama-tiny", "--steps", "400", "--seq-len", "32",
         "--grad-accum", "1", "--micro-batch", "1",
         "--ckpt-every", "100000", "--log-every", "50",
         "--ckpt-dir", str(tmp_path)],
        env=env, stdout=subprocess.PIPE, stderr=subprocess.STDOUT,
        text=True)
    try:
        time.sleep(4)                     # let it make a few steps
        p.send_signal(signal.SIGUSR1)
        deadline = time.monotonic() + 60
        while time.monotonic() < deadline:
            if any(n.startswith("ckpt_step")
                   for n in os.listdir(str(tmp_path))):
                break
            time.sleep(0.3)
        names = os.listdir(str(tmp_path))
        assert any(n.startswith("ckpt_step") for n in names), names
    finally:
        p.terminate()
        p.wait(timeout=60)


def test_fault_injection_knob_and_resume(tmp_path):
    """AITJ_FAULT_STEP kills the worker with the retryable code 137;
    a relaunch resumes from its checkpoint and completes."""
    import subprocess
    import sys
    env = {k: v for k, v in os.environ.items()
           if k not in ("RANK", "WORLD_SIZE")}
    env.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": "29751",
                "AITJ_FAULT_STEP": "3"})
    cmd = [sys.executable, "-m",
           "trainingjob_operator_amd.launcher.main",
           "--model", "llama-tiny", "--steps", "6", "--seq-len", "32",
           "--grad-accum", "1", "--micro-batch", "1",
           "--ckpt-every", "1", "--log-every", "1",
           "--ckpt-dir", str(tmp_path)]
    r = subprocess.run(cmd, env=env, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 137, (r.returncode, r.stderr[-500:])
    env.pop("AITJ_FAULT_STEP")
    r = subprocess.run(cmd, env=env, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, r.stderr[-1000:]
    assert "resumed from step" in r.stderr + r.stdout
