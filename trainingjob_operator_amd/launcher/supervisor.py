"""Local process supervisor: fault injection + restart-to-rejoin metric.

Mirrors what the operator does at cluster scope (restart scope All ->
delete + recreate the world) for a single node: spawn one launcher process
per rank, watch for failures, restart the whole world with an incremented
restart count, and measure the p50 restart-to-rejoin seconds — BASELINE.md
config 4's headline fault-tolerance metric — without needing a cluster.
Also drives elastic resize (config 3) by changing world size between
generations.
"""
from __future__ import annotations

import os
import signal
import subprocess
import sys
import threading
import time
from dataclasses import dataclass, field
from typing import Dict, List, Optional


@dataclass
class WorkerProc:
    rank: int
    proc: subprocess.Popen
    steps_seen: int = 0
    first_step_time: Optional[float] = None


@dataclass
class SupervisorReport:
    generations: int = 0
    restarts: int = 0
    rejoin_seconds: List[float] = field(default_factory=list)
    final_step: int = 0

    @property
    def p50_rejoin(self) -> Optional[float]:
        if not self.rejoin_seconds:
            return None
        s = sorted(self.rejoin_seconds)
        return s[len(s) // 2]


class LocalSupervisor:
    def __init__(self, world_size: int, ckpt_dir: str,
                 model: str = "llama-tiny", total_steps: int = 20,
                 seq_len: int = 32, master_port: int = 29765,
                 extra_args: Optional[List[str]] = None):
        self.world_size = world_size
        self.ckpt_dir = ckpt_dir
        self.model = model
        self.total_steps = total_steps
        self.seq_len = seq_len
        self.master_port = master_port
        self.extra_args = extra_args or []
        self.report = SupervisorReport()
        self.workers: Dict[int, WorkerProc] = {}
        self._restart_count = 0
        self._epoch = 0
        self._lock = threading.Lock()

    # -- process management ----------------------------------------------
    def _spawn(self, rank: int) -> WorkerProc:
        env = dict(os.environ)
        env.update({
            "RANK": str(rank),
            "WORLD_SIZE": str(self.world_size),
            "LOCAL_RANK": str(rank),
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(self.master_port),
            "TRAININGJOB_REPLICA_NAME": "trainer",
            "TRAININGJOB_REPLICA_INDEX": str(rank),
            "TRAININGJOB_REPLICA_RESTARTCOUNT": str(self._restart_count),
            "TRAININGJOB_RENDEZVOUS_EPOCH": str(self._epoch),
        })
        cmd = [sys.executable, "-m",
               "trainingjob_operator_amd.launcher.main",
               "--model", self.model, "--steps", str(self.total_steps),
               "--seq-len", str(self.seq_len), "--ckpt-dir", self.ckpt_dir,
               "--ckpt-every", "2", "--log-every", "1",
               "--grad-accum", "1", *self.extra_args]
        proc = subprocess.Popen(cmd, env=env, stdout=subprocess.PIPE,
                                stderr=subprocess.STDOUT, text=True)
        w = WorkerProc(rank, proc)
        threading.Thread(target=self._pump, args=(w,), daemon=True).start()
        return w

    def _pump(self, w: WorkerProc) -> None:
        for line in w.proc.stdout:
            if "step " in line and " loss " in line:
                with self._lock:
                    w.steps_seen += 1
                    if w.first_step_time is None:
                        w.first_step_time = time.monotonic()
                    try:
                        s = int(line.split("step ")[1].split()[0])
                        self.report.final_step = max(self.report.final_step, s)
                    except (ValueError, IndexError):
                        pass

    def start_world(self) -> None:
        self.report.generations += 1
        self.workers = {r: self._spawn(r) for r in range(self.world_size)}

    def kill_rank(self, rank: int) -> float:
        """SIGKILL one worker (the config-4 fault); returns the kill time."""
        self.workers[rank].proc.send_signal(signal.SIGKILL)
        return time.monotonic()

    def stop_world(self, sig=signal.SIGKILL) -> None:
        for w in self.workers.values():
            if w.proc.poll() is None:
                w.proc.send_signal(sig)
        for w in self.workers.values():
            try:
                w.proc.wait(timeout=30)
            except subprocess.TimeoutExpired:
                w.proc.kill()

    def restart_world(self, kill_time: float,
                      new_world_size: Optional[int] = None) -> None:
        """Operator restart semantics (scope All): tear down every rank,
        bump restart count (or rendezvous epoch on resize), respawn."""
        self.stop_world()
        if new_world_size is not None and new_world_size != self.world_size:
            self.world_size = new_world_size
            self._epoch += 1
        else:
            self._restart_count += 1
            self.report.restarts += 1
        self.start_world()
        # rejoin = fault -> the new world completes a step. Steps are
        # collective (DDP all-reduce), so rank 0 stepping implies every rank
        # rejoined; only rank 0 logs steps.
        deadline = time.monotonic() + 300
        while time.monotonic() < deadline:
            with self._lock:
                w0 = self.workers.get(0)
                if w0 is not None and w0.first_step_time is not None:
                    self.report.rejoin_seconds.append(
                        w0.first_step_time - kill_time)
                    return
            if all(w.proc.poll() is not None for w in self.workers.values()):
                break  # whole world exited (probably completed)
            time.sleep(0.1)

    def wait(self, timeout: float = 600) -> List[int]:
        deadline = time.monotonic() + timeout
        codes = []
        for w in self.workers.values():
            remaining = max(1.0, deadline - time.monotonic())
            try:
                codes.append(w.proc.wait(timeout=remaining))
            except subprocess.TimeoutExpired:
                w.proc.kill()
                codes.append(-9)
        return codes
