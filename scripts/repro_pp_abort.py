"""Bisect the PP llama-tiny SIGABRT: dense tiny trainer vs PP trainer."""
import faulthandler, os, sys
faulthandler.enable()
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

which = sys.argv[1]
from trainingjob_operator_amd.training import TrainConfig, Trainer

cfg = TrainConfig(model="llama-tiny", micro_batch=1, grad_accum=2,
                  seq_len=64, lr=1e-3)
if which == "dense":
    tr = Trainer(cfg, device=torch.device("cuda:0"))
else:
    from trainingjob_operator_amd.parallel.pp import PPTrainer
    tr = PPTrainer(cfg, stage_idx=0, n_stages=1, device="cuda:0")
for i in range(3):
    l = float(tr.train_step())
    torch.cuda.synchronize()
    print(f"{which} step {i} loss {l:.4f}", flush=True)
print(f"{which}: OK")
