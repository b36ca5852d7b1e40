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
