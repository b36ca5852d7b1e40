"""End-to-end DecoupledTrainer runs on CPU/gloo world_size=2 (BASELINE.json
config 1: GPT-2-small-shaped train=ddp on synthetic data; plus acco)."""

import os

import torch

from tests.conftest import run_distributed
from tests.dist_utils import teardown_worker


def _run_trainer(rank, world, port, tmpdir, method):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.chdir(tmpdir)

    from acco_amd.config import load_config
    from acco_amd.data.synthetic import SyntheticCausalLMDataset
    from acco_amd.engine.trainer import DecoupledTrainer
    from acco_amd.models import GPTNeoConfig, GPTNeoForCausalLM

    cfg = load_config([
        f"train={method}", "data=synthetic", "model=gptneo",
        "train.nb_steps_tot=8", "train.batch_size=2", "train.max_length=16",
        "train.use_mixed_precision=false", "train.save=false",
        "train.warmup=2", "train.n_warmup_steps=0",
        "train.dataloader_num_workers=0", "train.comm_buckets=2",
        "train.dataloader_persistent_workers=false",
    ])
    torch.manual_seed(42)
    mcfg = GPTNeoConfig(hidden_size=32, num_layers=2, num_heads=2,
                        vocab_size=64, max_position_embeddings=32,
                        window_size=8)
    model = GPTNeoForCausalLM(mcfg)
    train_ds = SyntheticCausalLMDataset(32, 16, 64, seed=5 + rank)

    trainer = DecoupledTrainer(model=model, train_dataset=train_ds,
                               eval_dataset=None, args=cfg.train,
                               run_name="itest")
    trainer.train()

    flat = trainer.params[:trainer.n_live].clone()
    torch.save({"params": flat}, os.path.join(tmpdir, f"p_{method}_{rank}.pt"))
    teardown_worker()


def _worker_ddp(rank, world, port, tmpdir):
    _run_trainer(rank, world, port, tmpdir, "ddp")


def _worker_acco(rank, world, port, tmpdir):
    _run_trainer(rank, world, port, tmpdir, "acco")


def _worker_dpu(rank, world, port, tmpdir):
    _run_trainer(rank, world, port, tmpdir, "dpu")


def _check(tmpdir, method, world=2):
    res = [torch.load(os.path.join(tmpdir, f"p_{method}_{r}.pt"),
                      weights_only=False) for r in range(world)]
    for r in range(1, world):
        assert torch.equal(res[0]["params"], res[r]["params"]), \
            f"{method}: rank {r} diverged"
    assert torch.isfinite(res[0]["params"]).all()
    # results.csv written by rank 0
    assert os.path.exists(os.path.join(tmpdir, "results.csv"))


def test_trainer_ddp_ws2():
    tmpdir = run_distributed(_worker_ddp, 2, timeout=300)
    _check(tmpdir, "ddp")


def test_trainer_acco_ws2():
    tmpdir = run_distributed(_worker_acco, 2, timeout=300)
    _check(tmpdir, "acco")


def test_trainer_acco_ws3_ragged():
    """world=3: the flat-arena shards do not divide evenly — the full
    trainer (engine + collectives + sharded AdamW) must still keep every
    rank's params bitwise identical."""
    tmpdir = run_distributed(_worker_acco, 3, timeout=300)
    _check(tmpdir, "acco", world=3)


def test_trainer_dpu_ws2():
    tmpdir = run_distributed(_worker_dpu, 2, timeout=300)
    _check(tmpdir, "dpu")


def _worker_acco_bf16(rank, world, port, tmpdir):
    """bf16 arenas + fp32 master shard on CPU (the mixed-precision data
    path of the GPU configuration, minus autocast)."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    os.chdir(tmpdir)

    from acco_amd.config import load_config
    from acco_amd.data.synthetic import SyntheticCausalLMDataset
    from acco_amd.engine.trainer import DecoupledTrainer
    from acco_amd.models import GPTNeoConfig, GPTNeoForCausalLM

    cfg = load_config([
        "train=acco", "train.nb_steps_tot=6", "train.batch_size=2",
        "train.max_length=16", "train.use_mixed_precision=true",
        "train.save=false", "train.dataloader_num_workers=0",
        "train.comm_buckets=2", "train.dataloader_persistent_workers=false",
    ])
    torch.manual_seed(42)
    mcfg = GPTNeoConfig(hidden_size=32, num_layers=1, num_heads=2,
                        vocab_size=64, max_position_embeddings=32,
                        window_size=8)
    model = GPTNeoForCausalLM(mcfg)
    ds = SyntheticCausalLMDataset(16, 16, 64, seed=5 + rank)
    trainer = DecoupledTrainer(model=model, train_dataset=ds,
                               eval_dataset=None, args=cfg.train,
                               run_name="bf16")
    assert trainer.params.dtype == torch.bfloat16
    assert trainer.opt.p.dtype == torch.float32
    trainer.train()
    flat = trainer.params[:trainer.n_live].float().clone()
    torch.save({"params": flat}, os.path.join(tmpdir, f"p_bf16_{rank}.pt"))
    teardown_worker()


def test_trainer_acco_bf16_ws2():
    tmpdir = run_distributed(_worker_acco_bf16, 2, timeout=300)
    res = [torch.load(os.path.join(tmpdir, f"p_bf16_{r}.pt"),
                      weights_only=False) for r in range(2)]
    assert torch.equal(res[0]["params"], res[1]["params"])
    assert torch.isfinite(res[0]["params"]).all()


class _RaggedDS(torch.utils.data.Dataset):
    def __init__(self, n, vocab, seed):
        self.n, self.vocab, self.seed = n, vocab, seed

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        g = torch.Generator().manual_seed(self.seed + i)
        L = int(torch.randint(4, 14, (1,), generator=g))
        return {"input_ids": torch.randint(0, self.vocab, (L,), generator=g)}


def _worker_acco_ft(rank, world, port, tmpdir):
    """Finetune-shaped path: ragged batches, padded collator, label
    masking (const_len_batch=false)."""
    os.environ.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
                       "RANK": str(rank), "LOCAL_RANK": str(rank),
                       "WORLD_SIZE": str(world)})
    os.chdir(tmpdir)
    from acco_amd.config import load_config
    from acco_amd.engine.trainer import DecoupledTrainer
    from acco_amd.models import GPTNeoConfig, GPTNeoForCausalLM

    cfg = load_config([
        "train=acco-ft", "train.nb_steps_tot=6", "train.batch_size=2",
        "train.use_mixed_precision=false", "train.save=false",
        "train.eval=false", "train.n_grad_accumulation=1",
        "train.dataloader_num_workers=0", "train.comm_buckets=2",
        "train.dataloader_persistent_workers=false",
    ])
    torch.manual_seed(42)
    mcfg = GPTNeoConfig(hidden_size=32, num_layers=1, num_heads=2,
                        vocab_size=64, max_position_embeddings=128,
                        window_size=8)
    model = GPTNeoForCausalLM(mcfg)

    class Tok:
        eos_token_id = 0

    trainer = DecoupledTrainer(model=model, tokenizer=Tok(),
                               train_dataset=_RaggedDS(32, 64, 5 + rank),
                               eval_dataset=None, args=cfg.train,
                               run_name="ft")
    trainer.train()
    torch.save({"params": trainer.params[:trainer.n_live].clone()},
               os.path.join(tmpdir, f"p_ft_{rank}.pt"))
    teardown_worker()


def test_trainer_acco_finetune_ragged_ws2():
    tmpdir = run_distributed(_worker_acco_ft, 2, timeout=300)
    res = [torch.load(os.path.join(tmpdir, f"p_ft_{r}.pt"),
                      weights_only=False) for r in range(2)]
    assert torch.equal(res[0]["params"], res[1]["params"])
    assert torch.isfinite(res[0]["params"]).all()


def _worker_eval(rank, world, port, tmpdir):
    """Eval cadence (reference eval_loop :399-415 + train_acco eval at
    :527-531): eval_loss computed every eval_step grads and logged."""
    os.environ.update({"MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
                       "RANK": str(rank), "LOCAL_RANK": str(rank),
                       "WORLD_SIZE": str(world)})
    os.chdir(tmpdir)
    from acco_amd.config import load_config
    from acco_amd.data.synthetic import SyntheticCausalLMDataset
    from acco_amd.engine.trainer import DecoupledTrainer
    from acco_amd.models import GPTNeoConfig, GPTNeoForCausalLM

    cfg = load_config([
        "train=acco", "train.nb_steps_tot=8", "train.batch_size=2",
        "train.max_length=16", "train.use_mixed_precision=false",
        "train.save=false", "train.eval=true", "train.eval_step=2",
        "train.n_grad_accumulation=1", "train.comm_buckets=2",
        "train.dataloader_num_workers=0",
        "train.dataloader_persistent_workers=false",
    ])
    torch.manual_seed(42)
    mcfg = GPTNeoConfig(hidden_size=32, num_layers=1, num_heads=2,
                        vocab_size=64, max_position_embeddings=64,
                        window_size=8)
    model = GPTNeoForCausalLM(mcfg)

    trainer = DecoupledTrainer(
        model=model, tokenizer=None,
        train_dataset=SyntheticCausalLMDataset(16, 16, 64, seed=3 + rank),
        eval_dataset=SyntheticCausalLMDataset(4, 16, 64, seed=99),
        args=cfg.train, run_name="evalrun")
    evals = []
    orig = trainer.eval_loop

    def spy():
        v = orig()
        evals.append(v)
        return v

    trainer.eval_loop = spy
    trainer.train()
    torch.save({"n_evals": len(evals), "vals": evals},
               os.path.join(tmpdir, f"ev_{rank}.pt"))
    teardown_worker()


def test_trainer_eval_cadence_ws2():
    tmpdir = run_distributed(_worker_eval, 2, timeout=300)
    res = torch.load(os.path.join(tmpdir, "ev_0.pt"), weights_only=False)
    assert res["n_evals"] >= 1          # cadence fired (rank 0 logs it)
    assert all(v == v and v > 0 for v in res["vals"])  # finite, positive


def test_trainer_ddp_ws3_ragged():
    tmpdir = run_distributed(_worker_ddp, 3, timeout=300)
    _check(tmpdir, "ddp", world=3)
