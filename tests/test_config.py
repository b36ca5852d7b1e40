from acco_amd.config import load_config


def test_defaults():
    cfg = load_config([])
    assert cfg.train.method_name == "acco"
    assert cfg.data.kind == "synthetic"
    assert cfg.model.family == "gptneo"
    assert cfg.train.batch_size == 8
    assert cfg.train.max_length == 1024


def test_group_selection_and_overrides():
    cfg = load_config(["train=ddp", "model=llama-1b",
                       "train.nb_steps_tot=123", "train.batch_size=2",
                       "run_name=x", "train.use_mixed_precision=false"])
    assert cfg.train.method_name == "ddp"
    assert cfg.train.run_baseline_ddp is True
    assert cfg.model.family == "llama"
    assert cfg.model.hidden_size == 2048
    assert cfg.train.nb_steps_tot == 123
    assert cfg.train.batch_size == 2
    assert cfg.run_name == "x"
    assert cfg.train.use_mixed_precision is False


def test_all_train_presets_load():
    for name in ["acco", "ddp", "dpu", "acco-ft", "ddp-ft", "dpu-ft"]:
        cfg = load_config([f"train={name}"])
        # full reference flag surface present (config/train/acco.yaml:1-28)
        for key in ["batch_size", "n_grad_accumulation", "learning_rate",
                    "weight_decay", "adam_beta1", "adam_beta2",
                    "nb_steps_tot", "label_smoothing_factor", "max_length",
                    "scheduler_name", "warmup", "use_mixed_precision",
                    "n_warmup_steps", "run_baseline_ddp", "method_name",
                    "eval", "save", "eval_step", "const_len_batch",
                    "finetune"]:
            assert key in cfg.train, (name, key)


def test_to_container_roundtrip():
    cfg = load_config([])
    d = cfg.to_container()
    assert isinstance(d, dict) and d["train"]["method_name"] == "acco"
