"""End-to-end checkpoint -> resume check (full-resume capability the
reference lacks): train N rounds, save, rebuild everything in
a fresh trainer, resume, train N more rounds. Asserts the restored trainer
reproduces the saved parameters exactly and keeps learning.

Runs on CPU (tiny model) or GPU (llama-1b shape with --big).
"""

import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch


def make_trainer(steps, seed):
    from acco_amd.config import load_config
    from acco_amd.data.synthetic import SyntheticCausalLMDataset
    from acco_amd.engine.trainer import DecoupledTrainer
    from acco_amd.models import (GPTNeoConfig, GPTNeoForCausalLM, LlamaConfig,
                                 LlamaForCausalLM)

    big = "--big" in sys.argv
    cfg = load_config([
        "train=acco", f"train.nb_steps_tot={steps}",
        "train.batch_size=4", f"train.max_length={256 if big else 32}",
        "train.save=false", "train.eval=false", "train.warmup=4",
        "train.n_grad_accumulation=1", "train.dataloader_num_workers=0",
        "train.dataloader_persistent_workers=false",
        "train.use_mixed_precision=" + ("true" if big else "false"),
    ])
    torch.manual_seed(seed)
    if big:
        mcfg = LlamaConfig(hidden_size=2048, num_layers=4, num_heads=32,
                           num_kv_heads=8, intermediate_size=8192,
                           vocab_size=50304, max_position_embeddings=4096)
        model = LlamaForCausalLM(mcfg)
    else:
        mcfg = GPTNeoConfig(hidden_size=64, num_layers=2, num_heads=2,
                            vocab_size=128, max_position_embeddings=64,
                            window_size=16)
        model = GPTNeoForCausalLM(mcfg)
    ds = SyntheticCausalLMDataset(64, cfg.train.max_length, mcfg.vocab_size,
                                  seed=9)
    return DecoupledTrainer(model=model, tokenizer=None, train_dataset=ds,
                            eval_dataset=None, args=cfg.train,
                            run_name="resume_check")


def main():
    path = os.path.join(tempfile.mkdtemp(prefix="acco_resume_"), "ck.pt")

    t1 = make_trainer(steps=10, seed=42)
    t1.train()
    t1.save_checkpoint(path)
    p_saved = t1.params[:t1.n_live].clone()
    loss_at_save = float(t1.engine.loss_static.item())

    # fresh process-equivalent: new model object, new trainer, restore
    t2 = make_trainer(steps=10, seed=777)   # different init on purpose
    t2.load_checkpoint(path)
    assert torch.equal(t2.params[:t2.n_live].cpu(), p_saved.cpu()), \
        "restored params differ from saved"
    assert t2.engine.count_grad_tot == t1.engine.count_grad_tot
    assert t2.sched.current_step == t1.sched.current_step

    t2.nb_grad_tot = t2.engine.count_grad_tot + 10
    t2.train()
    loss_after = float(t2.engine.loss_static.item())
    assert loss_after == loss_after, "NaN after resume"
    print(f"RESUME_OK saved_loss={loss_at_save:.4f} "
          f"resumed+10rounds_loss={loss_after:.4f} "
          f"grads={t2.engine.count_grad_tot}")


if __name__ == "__main__":
    main()
