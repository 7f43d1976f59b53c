"""Fault-tolerance and elastic-resize training tests (BASELINE.md configs
3-4) on CPU/gloo via the local supervisor: SIGKILL -> world restart ->
checkpoint-resume; resize 2->1 mid-run with the same checkpoints."""
import os
import signal
import time

import pytest

from trainingjob_operator_amd.launcher.supervisor import LocalSupervisor


@pytest.mark.timeout(600)
def test_sigkill_restart_resumes_from_checkpoint(tmp_path):
    sup = LocalSupervisor(world_size=2, ckpt_dir=str(tmp_path),
                          total_steps=12, master_port=29811)
    sup.start_world()
    # let it make progress past a checkpoint
    deadline = time.monotonic() + 120
    while sup.report.final_step < 4 and time.monotonic() < deadline:
        time.sleep(0.2)
    assert sup.report.final_step >= 4, "world never made progress"

    kill_t = sup.kill_rank(1)
    sup.restart_world(kill_t)
    codes = sup.wait(timeout=300)
    assert codes == [0, 0]
    assert sup.report.restarts == 1
    assert sup.report.p50_rejoin is not None
    assert sup.report.final_step >= 12
    # resumed from checkpoint, not from scratch: at least one ckpt file
    assert any(f.startswith("ckpt_step") for f in os.listdir(tmp_path))


@pytest.mark.timeout(600)
def test_elastic_resize_resumes_at_new_world(tmp_path):
    sup = LocalSupervisor(world_size=2, ckpt_dir=str(tmp_path),
                          total_steps=10, master_port=29833)
    sup.start_world()
    deadline = time.monotonic() + 120
    while sup.report.final_step < 3 and time.monotonic() < deadline:
        time.sleep(0.2)
    assert sup.report.final_step >= 3

    t = time.monotonic()
    sup.restart_world(t, new_world_size=1)  # scale 2 -> 1
    codes = sup.wait(timeout=300)
    assert codes == [0]
    assert sup.report.final_step >= 10
    assert sup._epoch == 1


@pytest.mark.timeout(600)
def test_fault_then_resize_sequence(tmp_path):
    """BASELINE configs 3+4 combined: train at world 2, SIGKILL -> restart,
    then elastic resize 2 -> 1, finishing from the same checkpoints."""
    sup = LocalSupervisor(world_size=2, ckpt_dir=str(tmp_path),
                          total_steps=14, master_port=29871)
    sup.start_world()
    deadline = time.monotonic() + 120
    while sup.report.final_step < 3 and time.monotonic() < deadline:
        time.sleep(0.2)
    assert sup.report.final_step >= 3

    # fault: kill rank 0, world restarts at same size
    kill_t = sup.kill_rank(0)
    sup.restart_world(kill_t)
    deadline = time.monotonic() + 120
    while sup.report.final_step < 7 and time.monotonic() < deadline:
        time.sleep(0.2)
    assert sup.report.restarts == 1
    assert sup.report.p50_rejoin is not None

    # elastic resize down to 1 and run to completion
    sup.restart_world(time.monotonic(), new_world_size=1)
    codes = sup.wait(timeout=300)
    assert codes == [0]
    assert sup.report.final_step >= 14
    assert sup._epoch == 1


@pytest.mark.timeout(600)
def test_elastic_resize_up(tmp_path):
    """Scale 1 -> 2 mid-run: the flat DP checkpoint resumes at the larger
    world (BASELINE config 3 upward direction)."""
    sup = LocalSupervisor(world_size=1, ckpt_dir=str(tmp_path),
                          total_steps=10, master_port=29877)
    sup.start_world()
    deadline = time.monotonic() + 120
    while sup.report.final_step < 3 and time.monotonic() < deadline:
        time.sleep(0.2)
    assert sup.report.final_step >= 3

    sup.restart_world(time.monotonic(), new_world_size=2)  # scale 1 -> 2
    codes = sup.wait(timeout=300)
    assert codes == [0, 0]
    assert sup.report.final_step >= 10
    assert sup._epoch == 1
