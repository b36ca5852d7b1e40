"""Checkpoint capabilities: HF-layout model state_dict (reference parity)
and full resume (model+optimizer+scheduler) — beyond-reference capability."""

import os
import tempfile

import torch

from acco_amd.engine.scheduler import LRSchedule
from acco_amd.engine.sharded_adamw import ShardedAdamW
from acco_amd.models import GPTNeoConfig, GPTNeoForCausalLM
from acco_amd.parallel.comm import ShardSpec


def test_model_state_dict_hf_layout():
    cfg = GPTNeoConfig(hidden_size=32, num_layers=2, num_heads=2,
                       vocab_size=64, max_position_embeddings=32)
    m = GPTNeoForCausalLM(cfg)
    keys = set(m.state_dict().keys())
    assert "transformer.wte.weight" in keys
    assert "transformer.h.0.attn.attention.q_proj.weight" in keys
    assert "transformer.h.1.mlp.c_fc.bias" in keys
    assert "transformer.ln_f.weight" in keys


def test_sharded_adamw_state_roundtrip():
    spec = ShardSpec.build(100, 2, buckets=2, align=4)
    dev = torch.device("cpu")
    a = ShardedAdamW(spec, 0, dev, lr=1e-3)
    a.p.normal_(); a.m.normal_(); a.v.uniform_(); a.step_count = 7
    b = ShardedAdamW(spec, 0, dev, lr=1e-3)
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "opt.pt")
        torch.save(a.state_dict(), path)
        b.load_state_dict(torch.load(path, weights_only=False))
    assert b.step_count == 7
    assert torch.equal(a.p, b.p) and torch.equal(a.m, b.m) and torch.equal(a.v, b.v)


def test_scheduler_state_roundtrip():
    s = LRSchedule(1.0, 10, 100)
    s.advance(17)
    t = LRSchedule(1.0, 10, 100)
    t.load_state_dict(s.state_dict())
    assert t.lr() == s.lr()


def _worker_consolidate(rank, world, port, tmpdir):
    import os
    import torch
    from tests.dist_utils import init_worker, teardown_worker
    init_worker(rank, world, port)
    spec = ShardSpec.build(40, world, buckets=2, align=4)
    opt = ShardedAdamW(spec, rank, torch.device("cpu"), lr=1e-3)
    # distinctive per-position values: owned segment holds its global index
    for j in range(spec.nb):
        seg0 = j * spec.bucket_elems + rank * spec.seg
        spec.owned_view(opt.p, j).copy_(
            torch.arange(seg0, seg0 + spec.seg, dtype=torch.float32))
    full = opt.consolidate(world)
    if rank == 0:
        assert torch.equal(full["p"],
                           torch.arange(spec.total, dtype=torch.float32))
        torch.save({"ok": True}, os.path.join(tmpdir, "ok.pt"))
    teardown_worker()


def test_consolidate_ws2():
    import os
    from tests.conftest import run_distributed
    tmpdir = run_distributed(_worker_consolidate, 2, timeout=120)
    assert os.path.exists(os.path.join(tmpdir, "ok.pt"))


def test_load_pretrained_file_and_dir(tmp_path):
    """Finetune entry (reference main.py:33-41 capability): load weights
    from a state-dict file and from an HF-style model directory."""
    import torch

    from acco_amd.models import (LlamaConfig, LlamaForCausalLM,
                                 load_pretrained)

    cfg = LlamaConfig(hidden_size=32, num_layers=2, num_heads=4,
                      num_kv_heads=2, intermediate_size=64, vocab_size=64,
                      max_position_embeddings=64)
    torch.manual_seed(0)
    src = LlamaForCausalLM(cfg)

    f = tmp_path / "model.pt"
    torch.save(src.state_dict(), f)
    m1 = LlamaForCausalLM(cfg)
    load_pretrained(m1, str(f))
    for a, b in zip(src.parameters(), m1.parameters()):
        assert torch.equal(a, b)

    d = tmp_path / "hf_dir"
    d.mkdir()
    torch.save(src.state_dict(), d / "pytorch_model.bin")
    m2 = LlamaForCausalLM(cfg)
    load_pretrained(m2, str(d))
    assert torch.equal(m2.lm_head.weight, src.lm_head.weight)


def test_resume_end_to_end_cpu():
    """Full train -> save -> fresh trainer -> restore -> continue cycle
    (the scripted GPU check benchmarks/resume_check.py, CPU mode)."""
    import subprocess
    import sys
    import tempfile

    out = subprocess.run(
        [sys.executable,
         os.path.join(os.path.dirname(os.path.dirname(
             os.path.abspath(__file__))), "benchmarks", "resume_check.py")],
        capture_output=True, text=True, timeout=300,
        cwd=tempfile.mkdtemp(prefix="acco_resume_test_"))
    assert out.returncode == 0, out.stderr[-2000:]
    assert "RESUME_OK" in out.stdout, out.stdout[-500:]
