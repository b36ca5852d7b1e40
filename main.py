"""Training entry point.

CLI parity with the reference (reference main.py:25-67, README.md:52-58):

    python main.py train=acco data=synthetic model=gptneo \
        train.nb_steps_tot=100 run_name=myrun

Config groups and flag surface mirror the reference's Hydra setup; the data
default is `synthetic` because this environment has no network (use
data=openwebtext / data=alpaca on a machine with HF hub access).
"""

import logging
import os
import sys

import torch

from acco_amd.config import load_config
from acco_amd.data.synthetic import SyntheticCausalLMDataset
from acco_amd.engine.trainer import DecoupledTrainer
from acco_amd.models import build_model

logging.basicConfig(stream=sys.stdout, level=logging.INFO)
logger = logging.getLogger("acco_amd")


def build_datasets(cfg, model):
    data = cfg.data
    if data.get("kind") == "synthetic":
        vocab = model.cfg.vocab_size
        seq = cfg.train.max_length
        train = SyntheticCausalLMDataset(data.n_train_sequences, seq, vocab,
                                         seed=cfg.seed)
        evald = (SyntheticCausalLMDataset(data.n_eval_sequences, seq, vocab,
                                          seed=cfg.seed + 1)
                 if data.n_eval_sequences else None)
        return train, evald, None
    # HF path (hub, or a committed local corpus via kind=hf_local): same
    # split policy as the reference (train_test_split 0.05 seed 42,
    # reference main.py:50)
    from transformers import AutoTokenizer
    from acco_amd.data import load_raw_dataset, tokenizer_path
    ds = load_raw_dataset(data)
    ds = ds["train"].train_test_split(0.05, seed=42)
    tokenizer = AutoTokenizer.from_pretrained(tokenizer_path(cfg))
    tokenizer.pad_token_id = tokenizer.eos_token_id
    return ds["train"], ds["test"], tokenizer


def _resolve_run_dir(template: str) -> str:
    """Hydra-style date templating (reference config/config.yaml:10-12):
    ${now:%Y-%m-%d} etc. expand against the launch time."""
    import datetime
    import re
    now = datetime.datetime.now()
    return re.sub(r"\$\{now:([^}]*)\}", lambda m: now.strftime(m.group(1)),
                  template)


def _absolutize_local_paths(cfg) -> None:
    """Run dirs chdir (Hydra semantics); local corpus/tokenizer paths in the
    data group must survive that."""
    for key in ("path", "tokenizer"):
        v = cfg.data.get(key)
        if isinstance(v, str) and os.path.exists(v):
            cfg.data[key] = os.path.abspath(v)


def _enter_run_dir(cfg, job_subdir=None):
    """Create + chdir into the templated run dir (the reference's Hydra
    run-management: every run writes results.csv / checkpoints / scalars
    into its own date-stamped directory). run_dir=. disables."""
    template = cfg.get("run_dir") or "."
    if template in (".", "none", ""):
        return None
    d = _resolve_run_dir(template)
    if job_subdir is not None:
        d = os.path.join(d, str(job_subdir))
    os.makedirs(d, exist_ok=True)
    _absolutize_local_paths(cfg)
    prev = os.getcwd()
    os.chdir(d)
    logger.info("run dir: %s", os.getcwd())
    return prev


def _expand_multirun(overrides):
    """Hydra -m sweep: `a=1,2 b=x,y` → the cross product of run configs."""
    import itertools
    fixed, sweeps = [], []
    for ov in overrides:
        key, _, val = ov.partition("=")
        if "," in val and not val.startswith(("[", "{")):
            sweeps.append([(key, v) for v in val.split(",")])
        else:
            fixed.append(ov)
    if not sweeps:
        return [list(overrides)]
    return [fixed + [f"{k}={v}" for k, v in combo]
            for combo in itertools.product(*sweeps)]


def run_one(overrides, job_subdir=None):
    cfg = load_config(overrides)
    prev = _enter_run_dir(cfg, job_subdir)
    try:
        _train(cfg)
    finally:
        if prev is not None:
            os.chdir(prev)


def main(argv=None):
    argv = list(argv if argv is not None else sys.argv[1:])
    multirun = False
    for flag in ("-m", "--multirun"):
        if flag in argv:
            argv.remove(flag)
            multirun = True
    if multirun:
        jobs = _expand_multirun(argv)
        logger.info("multirun: %d jobs", len(jobs))
        for i, job in enumerate(jobs):
            logger.info("=== multirun job %d: %s", i, job)
            run_one(job, job_subdir=i)
        return
    run_one(argv)


def _train(cfg):
    torch.manual_seed(42)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(42)  # reference main.py:28

    model = build_model(cfg.model)
    if cfg.train.get("finetune"):
        # reference main.py:33-41: finetune starts from pretrained weights.
        # No-network environment: the checkpoint must be local —
        # model.pretrained_path (config) or train.pretrained_path override.
        path = (cfg.model.get("pretrained_path")
                or cfg.train.get("pretrained_path"))
        if path:
            from acco_amd.models import load_pretrained
            load_pretrained(model, path)
            logger.info("loaded pretrained weights from %s", path)
        else:
            logger.warning("finetune=true but no pretrained_path set; "
                           "starting from random init")
    logger.info("model instantiated (%.1fM params)",
                sum(p.numel() for p in model.parameters()) / 1e6)
    train_ds, eval_ds, tokenizer = build_datasets(cfg, model)

    trainer = DecoupledTrainer(
        model=model,
        tokenizer=tokenizer,
        train_dataset=train_ds,
        eval_dataset=eval_ds,
        text_column_name="text",
        args=cfg.train,
        log=logger,
        preprocess_dataset_fn=None,
        run_name=cfg.run_name,
    )
    trainer.train()


if __name__ == "__main__":
    main()
