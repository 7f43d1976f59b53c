#!/usr/bin/env python3
"""Single-rank MoE (EPTrainer) step bench — for timing + rocprofv3 stats."""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def main():
    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 5
    model = sys.argv[2] if len(sys.argv) > 2 else "moe-mid"
    from trainingjob_operator_amd.parallel.ep import EPTrainer
    from trainingjob_operator_amd.training import TrainConfig
    cfg = TrainConfig(model=model, micro_batch=1, grad_accum=1,
                      seq_len=4096, lr=1e-4)
    tr = EPTrainer(cfg, device="cuda:0")
    for _ in range(2):
        tr.train_step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        loss = tr.train_step()
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / steps * 1e3
    tok = cfg.micro_batch * cfg.seq_len / (ms / 1e3)
    print(f"{model}: {ms:.1f} ms/step  {tok:.0f} tok/s  loss={float(loss):.3f}")


if __name__ == "__main__":
    main()
