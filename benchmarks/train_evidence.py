"""End-to-end training evidence on MI355X: GPT-Neo-125M learns a synthetic
successor task with ALL custom HIP kernels active; prints the loss curve."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
from torch.utils.data import Dataset


class SuccessorDataset(Dataset):
    def __init__(self, n, seq, vocab, seed):
        self.n, self.seq, self.vocab, self.seed = n, seq, vocab, seed

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        g = torch.Generator().manual_seed(self.seed + i)
        start = torch.randint(0, self.vocab, (1,), generator=g)
        ids = (start + torch.arange(self.seq)) % self.vocab
        return {"input_ids": ids}


def main():
    from acco_amd.config import load_config
    from acco_amd.engine.trainer import DecoupledTrainer
    from acco_amd.models import GPTNeoConfig, GPTNeoForCausalLM

    steps = int(sys.argv[1]) if len(sys.argv) > 1 else 200
    use_llama = "--llama" in sys.argv
    cfg = load_config([
        "train=acco", f"train.nb_steps_tot={steps}", "train.batch_size=8",
        "train.max_length=256", "train.save=false", "train.warmup=20",
        "train.learning_rate=1.5e-3", "train.dataloader_num_workers=0",
        "train.dataloader_persistent_workers=false",
    ])
    torch.manual_seed(42)
    if use_llama:
        from acco_amd.models import LlamaConfig, LlamaForCausalLM
        mcfg = LlamaConfig(hidden_size=2048, num_layers=16, num_heads=32,
                           num_kv_heads=8, intermediate_size=8192,
                           vocab_size=50304, max_position_embeddings=4096)
        model = LlamaForCausalLM(mcfg)
    else:
        mcfg = GPTNeoConfig(hidden_size=768, num_layers=12, num_heads=12,
                            vocab_size=50304, max_position_embeddings=2048,
                            window_size=256)
        model = GPTNeoForCausalLM(mcfg)
    ds = SuccessorDataset(2048, 256, 50304, seed=7)

    losses = []
    trainer = DecoupledTrainer(model=model, train_dataset=ds,
                               eval_dataset=None, args=cfg.train,
                               run_name="evidence")
    orig = trainer.engine.on_round_complete

    def hook(round_idx, count):
        losses.append((count, float(trainer.engine.loss_static.item())))
        if orig:
            orig(round_idx, count)

    trainer.engine.on_round_complete = hook
    trainer.train()
    print("LOSS_CURVE", [(c, round(l, 3)) for c, l in losses[::5]])
    first = sum(l for _, l in losses[:5]) / 5
    last = sum(l for _, l in losses[-5:]) / 5
    print(f"first5={first:.3f} last5={last:.3f}")
    assert last < first - 1.0, "training did not learn (expect >1 nat drop)"
    print("TRAIN_EVIDENCE_OK")


if __name__ == "__main__":
    main()
