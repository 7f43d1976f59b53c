"""Token-file dataset (launcher/data.py): memory-mapped flat binaries,
deterministic per-rank sampling, Trainer integration."""
import os

import numpy as np
import pytest
import torch

from trainingjob_operator_amd.launcher.data import (
    TokenFileDataset, make_batches, write_token_file,
)
from trainingjob_operator_amd.training import TrainConfig, Trainer


@pytest.fixture
def token_file(tmp_path):
    rng = np.random.default_rng(3)
    path = os.path.join(str(tmp_path), "corpus.bin")
    return write_token_file(path, rng.integers(0, 512, size=5000))


def test_shapes_and_shift(token_file):
    ds = TokenFileDataset(token_file, seq_len=32, micro_batch=4)
    tokens, targets = next(ds.batches())
    assert tokens.shape == (4, 32) and targets.shape == (4, 32)
    assert tokens.dtype == torch.int64
    # targets are the next-token shift of the same window
    assert torch.equal(tokens[:, 1:], targets[:, :-1])
    assert int(tokens.max()) < 512


def test_deterministic_and_rank_disjoint(token_file):
    a1 = next(TokenFileDataset(token_file, 32, 2, rank=0).batches())
    a2 = next(TokenFileDataset(token_file, 32, 2, rank=0).batches())
    b = next(TokenFileDataset(token_file, 32, 2, rank=1).batches())
    assert torch.equal(a1[0], a2[0])          # reproducible
    assert not torch.equal(a1[0], b[0])       # per-rank streams differ


def test_uint32_and_too_short(tmp_path, token_file):
    p32 = write_token_file(os.path.join(str(tmp_path), "c32.bin"),
                           np.arange(100) + 65536, dtype="uint32")
    tokens, _ = next(TokenFileDataset(p32, 16, 1, dtype="uint32").batches())
    assert int(tokens.max()) > 65535           # uint16 would have wrapped
    with pytest.raises(ValueError):
        TokenFileDataset(p32, seq_len=200, micro_batch=1, dtype="uint32")
    with pytest.raises(ValueError):
        TokenFileDataset(token_file, 16, 1, dtype="float64")


def test_trainer_trains_on_token_file(token_file):
    cfg = TrainConfig(model="llama-tiny", micro_batch=2, grad_accum=2,
                      seq_len=32, lr=2e-3, data_path=token_file)
    tr = Trainer(cfg)
    losses = [float(tr.train_step()) for _ in range(5)]
    assert all(l == l for l in losses)
    assert losses[-1] < losses[0] + 0.5        # sane trajectory


def test_make_batches_dispatch(token_file):
    cfg = TrainConfig(model="llama-tiny", seq_len=16, micro_batch=1)
    t_syn, _ = next(make_batches(cfg, None))
    cfg.data_path = token_file
    t_file, _ = next(make_batches(cfg, None))
    assert t_syn.shape == t_file.shape
    cfg.data_path = "/nonexistent/corpus.bin"
    with pytest.raises(FileNotFoundError):
        make_batches(cfg, None)


def _pp_data_worker(rank, world, port, data_path, ckdir):
    import torch.distributed  # noqa: F401
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "WORLD_SIZE": str(world),
        "LOCAL_RANK": str(rank),
    })
    from trainingjob_operator_amd.launcher.main import main
    rc = main(["--model", "llama-tiny", "--steps", "2", "--seq-len", "32",
               "--grad-accum", "1", "--micro-batch", "1",
               "--ckpt-every", "10", "--log-every", "1",
               "--ckpt-dir", ckdir, "--pp", str(world),
               "--data-path", data_path])
    assert rc == 0


def test_pp_launcher_trains_on_token_file(token_file, tmp_path):
    """--pp with --data-path: pipeline stages draw the same real-data
    stream (stage 0 consumes tokens, the last consumes targets)."""
    import torch.multiprocessing as mp

    def _free_port():
        import socket
        s = socket.socket()
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
        s.close()
        return port

    mp.spawn(_pp_data_worker,
             args=(2, _free_port(), token_file, str(tmp_path)),
             nprocs=2, join=True)
