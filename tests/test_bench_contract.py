"""bench.py driver contract: one JSON line on stdout with the required
schema (the round driver parses exactly this)."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {"metric": str, "value": float, "unit": str, "n_gpus": int,
            "steps": int, "warmup": int, "ms_per_step": float,
            "higher_is_better": bool, "scaling": str, "dtype": str,
            "data": str, "config": dict}


def _run(args):
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py")] + args,
        capture_output=True, text=True, timeout=600, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    return json.loads(lines[0])


def test_bench_json_contract_acco():
    d = _run(["--steps", "2", "--warmup", "1"])
    for k, t in REQUIRED.items():
        assert k in d, f"missing {k}"
        assert isinstance(d[k], t), (k, type(d[k]))
    assert "vs_baseline" in d            # null allowed (no published baseline)
    assert d["metric"] == "tokens/s"
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["data"] == "synthetic"
    cfg = d["config"]
    for k in ("model", "method", "global_batch", "seq_len", "parallelism"):
        assert k in cfg


def test_bench_json_contract_ddp():
    d = _run(["--method", "ddp", "--steps", "2", "--warmup", "1"])
    assert d["config"]["method"] == "ddp"
    assert d["value"] > 0


def test_bench_json_contract_dpu():
    d = _run(["--method", "dpu", "--steps", "2", "--warmup", "1"])
    assert d["config"]["method"] == "dpu"
    assert d["value"] > 0


def test_bench_multirank_launch_contract():
    """The driver's N>1 launch shape: torch.distributed.run --nnodes=1
    --nproc-per-node 2 bench.py --gpus 2 ... (gloo on CPU here; RCCL on a
    GPU node). Rank 0 must emit exactly one whole-job JSON line with
    n_gpus=2 and the SUM-over-ranks token throughput."""
    env = dict(os.environ)
    env.pop("RANK", None); env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29611", os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--steps", "2", "--warmup", "1"],
        capture_output=True, text=True, timeout=600, cwd=REPO, env=env)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["global_batch"] == 2 * 8
    assert d["value"] > 0


def test_bench_refuses_mislabeled_gpus_flag():
    """--gpus that contradicts WORLD_SIZE must refuse (exit 2), not emit a
    mislabeled record."""
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--gpus", "4",
         "--steps", "1", "--warmup", "1"],
        capture_output=True, text=True, timeout=300, cwd=REPO)
    assert out.returncode == 2
    assert not [l for l in out.stdout.splitlines() if l.startswith("{")]


def test_bench_json_contract_ckpt_flag():
    """--ckpt (activation checkpointing) runs and is recorded in config."""
    d = _run(["--steps", "2", "--warmup", "1", "--ckpt"])
    assert d["config"]["ckpt"] is True
    assert d["value"] > 0
