"""Results/observability helpers: results.csv schema-union merge (reference
utils/logs_utils.py:71-138), run-row construction, com-log summary tool."""

import csv
import json
import os
import subprocess
import sys

from acco_amd.utils.logging import create_dict_result, save_result

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_save_result_schema_union(tmp_path):
    p = str(tmp_path / "results.csv")
    save_result(p, {"run": "a", "loss": 1.0})
    save_result(p, {"run": "b", "tokens_s": 5.0})      # new column appears
    with open(p, newline="") as f:
        rows = list(csv.DictReader(f))
    assert len(rows) == 2
    assert set(rows[0].keys()) == {"run", "loss", "tokens_s"}
    assert rows[0]["tokens_s"] == ""                   # backfilled empty
    assert rows[1]["loss"] == "" and rows[1]["tokens_s"] == "5.0"


def test_create_dict_result_fields():
    row = create_dict_result({"learning_rate": 1e-4, "method_name": "acco"},
                             world_size=8, n_nodes=1, device_name="MI355X",
                             total_time=12.5, id_run="xyz", loss_final=2.25)
    assert row["N_workers"] == 8
    assert row["cuda_device"] == "MI355X"
    assert row["Tot_time"] == 12.5
    assert row["Loss_final"] == 2.25
    assert row["learning_rate"] == 1e-4


def test_com_log_summary_tool(tmp_path):
    p = str(tmp_path / "com_logs_x.json")
    json.dump([{"round": i, "commit": i % 2 == 1, "t": 0.01, "count": 2}
               for i in range(6)], open(p, "w"))
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "tools", "com_log_summary.py"), p],
        capture_output=True, text=True, timeout=60)
    assert out.returncode == 0
    assert "commit (odd)" in out.stdout and "total com-round wall time" in out.stdout
