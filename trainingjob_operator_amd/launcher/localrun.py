"""Local pod runner ("mini-kubelet") + the operator-path restart benchmark.

``LocalKubelet`` runs an AITrainingJob's pods as real launcher
subprocesses against a FakeKubeApi: it spawns a process for every pod the
controller creates (with EXACTLY the env the controller injected, the
headless-service DNS rewritten to loopback), reports phases back
(Running / Failed-with-exit-code / Succeeded), and SIGTERMs processes
whose pod the controller deleted. With a controller loop driving
``sync_once`` this closes the full operator <-> kubelet <-> worker cycle
that the reference delegates to a real cluster.

``restart_benchmark`` measures BASELINE config 4 *through the operator*:
SIGKILL one worker -> pod Failed(137) -> controller restart dance
(Terminating -> wait-gate -> Restarting -> recreate) -> new world loads
the checkpoint -> first post-rejoin checkpoint lands. This is the
controller-path complement to the launcher-supervisor number
(launcher/supervisor.py), with pod lifecycle and the two-sync dance in
the measured window.
"""
from __future__ import annotations

import os
import signal
import subprocess
import sys
import threading
import time
from typing import Dict, List, Optional

from ..api import constants as C


def free_port() -> int:
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


class LocalKubelet:
    def __init__(self, api, namespace: str, launcher_args: List[str],
                 master_port: int, poll_s: float = 0.02):
        self.api = api
        self.ns = namespace
        self.args = launcher_args
        self.port = master_port
        self.poll_s = poll_s
        self.procs: Dict[str, subprocess.Popen] = {}   # uid -> proc
        self.uid_name: Dict[str, str] = {}
        # uid -> monotonic time rank 0 logged its first completed step
        # (the supervisor's rejoin marker: steps are collective, so rank 0
        # stepping implies every rank rejoined)
        self.first_step: Dict[str, Optional[float]] = {}
        self.tail: Dict[str, list] = {}      # uid -> last output lines
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._lock = threading.Lock()

    # -- lifecycle ------------------------------------------------------
    def start(self) -> None:
        self._thread = threading.Thread(target=self._loop, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=10)
        with self._lock:
            for p in self.procs.values():
                if p.poll() is None:
                    p.terminate()
            for p in self.procs.values():
                try:
                    p.wait(timeout=30)
                except subprocess.TimeoutExpired:
                    p.kill()
            self.procs.clear()

    def kill_pod_process(self, name: str, sig=signal.SIGKILL) -> bool:
        """Fault injection: SIGKILL the process backing a pod (the pod
        object stays until the kubelet loop reports Failed)."""
        with self._lock:
            for uid, pname in self.uid_name.items():
                p = self.procs.get(uid)
                if pname == name and p is not None and p.poll() is None:
                    p.send_signal(sig)
                    return True
        return False

    # -- the kubelet loop ----------------------------------------------
    def _spawn(self, pod: dict) -> subprocess.Popen:
        env = {e["name"]: e.get("value", "")
               for e in pod["spec"]["containers"][0].get("env", [])}
        full_env = {**os.environ, **env,
                    "MASTER_ADDR": "127.0.0.1",
                    "MASTER_PORT": str(self.port)}
        full_env.pop("CUDA_VISIBLE_DEVICES", None)
        is_rank0 = pod["metadata"]["name"].endswith("-0")
        proc = subprocess.Popen(
            [sys.executable, "-m",
             "trainingjob_operator_amd.launcher.main"] + self.args,
            env=full_env,
            stdout=subprocess.PIPE if is_rank0 else subprocess.DEVNULL,
            stderr=subprocess.STDOUT, text=is_rank0)
        if is_rank0:
            uid = pod["metadata"]["uid"]

            tl = self.tail.setdefault(uid, [])

            def reader():
                for line in proc.stdout:
                    tl.append(line.rstrip())
                    del tl[:-60]
                    if " step " in line and " loss " in line and \
                            self.first_step.get(uid) is None:
                        self.first_step[uid] = time.monotonic()
                proc.stdout.close()

            threading.Thread(target=reader, daemon=True).start()
        return proc

    def _loop(self) -> None:
        while not self._stop.is_set():
            try:
                pods = {p["metadata"]["uid"]: p
                        for p in self.api.list_pods(self.ns)}
            except Exception:
                time.sleep(self.poll_s)
                continue
            with self._lock:
                # reap exits -> report phase once
                for uid, proc in list(self.procs.items()):
                    rc = proc.poll()
                    if rc is None:
                        continue
                    name = self.uid_name.get(uid, "")
                    if uid in pods:
                        try:
                            self.api.set_pod_phase(
                                self.ns, name,
                                "Succeeded" if rc == 0 else "Failed",
                                exit_code=(None if rc == 0 else
                                           (128 - rc if rc < 0 else rc)))
                        except Exception:
                            pass
                    del self.procs[uid]
                # SIGTERM processes whose pod the controller deleted
                for uid in list(self.procs):
                    if uid not in pods:
                        self.procs[uid].terminate()
                # spawn for new pods (once per pod UID)
                for uid, pod in pods.items():
                    if uid in self.procs or uid in self.uid_name:
                        continue
                    self.first_step[uid] = None
                    self.procs[uid] = self._spawn(pod)
                    self.uid_name[uid] = pod["metadata"]["name"]
                    try:
                        self.api.set_pod_phase(
                            self.ns, pod["metadata"]["name"], "Running")
                    except Exception:
                        pass
            time.sleep(self.poll_s)


def _latest_step(ckdir: str) -> int:
    try:
        steps = [int(n[len("ckpt_step"):-3]) for n in os.listdir(ckdir)
                 if n.startswith("ckpt_step") and n.endswith(".pt")]
        return max(steps) if steps else -1
    except (OSError, ValueError):
        return -1


def restart_benchmark(model: str = "llama-tiny", replicas: int = 2,
                      trials: int = 3, seq_len: int = 32,
                      ckpt_root: Optional[str] = None,
                      steps: int = 10_000, ckpt_every: int = 1,
                      sync_period_s: float = 0.05,
                      timeout_s: float = 300.0) -> List[float]:
    """Measured: SIGKILL(worker 0) -> [operator restart dance + respawn +
    checkpoint resume] -> first NEW checkpoint from the rejoined world.
    Returns per-trial seconds."""
    import tempfile

    from ..controller.core import TrainingJobController
    from ..controller.options import OperatorOptions
    from ..kube.fake import FakeKubeApi

    ns = "default"
    times: List[float] = []
    for trial in range(trials):
        root = ckpt_root or tempfile.mkdtemp(prefix="aitj-restart-")
        ckdir = os.path.join(root, f"trial{trial}")
        api = FakeKubeApi()
        tc = TrainingJobController(api, OperatorOptions())
        api.create_job(ns, {
            "apiVersion": C.API_VERSION, "kind": C.CRD_KIND,
            "metadata": {"name": "rb", "namespace": ns},
            "spec": {
                "restartingExitCode": "137,9",
                "replicaSpecs": {"trainer": {
                    "replicas": replicas,
                    "restartPolicy": "ExitCode", "restartScope": "All",
                    "restartLimit": 10,
                    "template": {"spec": {"containers": [{
                        "name": "aitj-trainer",
                        "ports": [{"name": "aitj-rccl",
                                   "containerPort": 23456}],
                    }]}},
                }},
            },
        })

        stop = threading.Event()

        def controller_loop():
            while not stop.is_set():
                try:
                    tc.sync_once(f"{ns}/rb")
                except Exception:
                    pass
                time.sleep(sync_period_s)

        ct = threading.Thread(target=controller_loop, daemon=True)
        ct.start()
        kubelet = LocalKubelet(api, ns, [
            "--model", model, "--steps", str(steps),
            "--seq-len", str(seq_len), "--grad-accum", "1",
            "--micro-batch", "1", "--ckpt-every", str(ckpt_every),
            "--log-every", "1",
            "--ckpt-dir", ckdir,
        ], free_port())
        kubelet.start()
        try:
            # wait for steady-state progress
            deadline = time.monotonic() + timeout_s
            while _latest_step(ckdir) < max(3, ckpt_every):
                if time.monotonic() > deadline:
                    for uid, lines in kubelet.tail.items():
                        print(f"--- rank0 tail ({kubelet.uid_name.get(uid)})")
                        print("\n".join(lines[-40:]))
                    raise TimeoutError("no initial progress")
                time.sleep(0.02)
            rank0_uids = {u for u, n in kubelet.uid_name.items()
                          if n == "rb-trainer-0"}
            t0 = time.monotonic()
            assert kubelet.kill_pod_process("rb-trainer-0")
            # rejoin marker (same as supervisor.py): the RESTARTED world's
            # rank 0 logs its first completed step
            deadline = time.monotonic() + timeout_s
            while True:
                done = [t for u, t in kubelet.first_step.items()
                        if u not in rank0_uids and t is not None
                        and kubelet.uid_name.get(u) == "rb-trainer-0"]
                if done:
                    times.append(done[0] - t0)
                    break
                if time.monotonic() > deadline:
                    raise TimeoutError("world never rejoined")
                time.sleep(0.005)
        finally:
            stop.set()
            ct.join(timeout=10)
            kubelet.stop()
            import shutil
            shutil.rmtree(ckdir, ignore_errors=True)   # 1B ckpts are ~17 GB
    return times


if __name__ == "__main__":
    import argparse
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="llama-tiny")
    ap.add_argument("--trials", type=int, default=3)
    ap.add_argument("--seq-len", type=int, default=32)
    ap.add_argument("--replicas", type=int, default=2)
    ap.add_argument("--ckpt-every", type=int, default=1)
    a = ap.parse_args()
    ts = restart_benchmark(model=a.model, trials=a.trials,
                           seq_len=a.seq_len, replicas=a.replicas,
                           ckpt_every=a.ckpt_every)
    ts_s = sorted(ts)
    print({"trials": [round(t, 3) for t in ts],
           "p50_s": round(ts_s[len(ts_s) // 2], 3)})
