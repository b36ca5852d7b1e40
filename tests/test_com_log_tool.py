"""tools/com_log_summary.py on a real com-log produced by a trainer run
(the reference's save_com_logs channel equivalent)."""

import glob
import json
import os

import torch

from tests.conftest import run_distributed
from tests.dist_utils import teardown_worker


def _worker(rank, world, port, tmpdir):
    os.environ.update({
        "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
        "RANK": str(rank), "LOCAL_RANK": str(rank),
        "WORLD_SIZE": str(world),
    })
    os.chdir(tmpdir)
    from acco_amd.config import load_config
    from acco_amd.data.synthetic import SyntheticCausalLMDataset
    from acco_amd.engine.trainer import DecoupledTrainer
    from acco_amd.models import GPTNeoConfig, GPTNeoForCausalLM

    cfg = load_config([
        "train=acco", "data=synthetic", "model=gptneo",
        "train.nb_steps_tot=6", "train.batch_size=2", "train.max_length=16",
        "train.use_mixed_precision=false", "train.save=false",
        "train.n_warmup_steps=0", "train.dataloader_num_workers=0",
        "train.comm_buckets=2",
        "train.dataloader_persistent_workers=false",
    ])
    torch.manual_seed(42)
    mcfg = GPTNeoConfig(hidden_size=32, num_layers=1, num_heads=2,
                        vocab_size=64, max_position_embeddings=32,
                        window_size=8)
    trainer = DecoupledTrainer(model=GPTNeoForCausalLM(mcfg),
                               train_dataset=SyntheticCausalLMDataset(
                                   16, 16, 64, seed=3 + rank),
                               eval_dataset=None, args=cfg.train,
                               run_name="comlog")
    trainer.train()
    teardown_worker()


def test_com_log_dump_and_summary(capsys):
    tmpdir = run_distributed(_worker, 2, timeout=240)
    logs = glob.glob(os.path.join(tmpdir, "com_logs_*.json"))
    assert logs, "rank 0 must dump the com log at end of training"
    rounds = json.load(open(logs[0]))
    assert rounds and {"round", "commit", "t", "count"} <= set(rounds[0])
    assert any(r["commit"] for r in rounds)
    assert any(not r["commit"] for r in rounds)

    import importlib.util
    spec = importlib.util.spec_from_file_location(
        "com_log_summary",
        os.path.join(os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))), "tools", "com_log_summary.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    mod.summarize(logs[0])
    out = capsys.readouterr().out
    assert "commit (odd)" in out and "total com-round wall time" in out
