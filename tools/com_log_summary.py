"""Summarize a com_logs_<id>.json dump (written by DecoupledTrainer at end
of training — the reference's save_com_logs channel,
utils/logs_utils.py:141-152): per-round communication wall time and grad
counts, split by tentative/commit parity.

Usage: python tools/com_log_summary.py com_logs_XXXX.json
"""

import json
import statistics
import sys


def summarize(path: str) -> None:
    rounds = json.load(open(path))
    if not rounds:
        print("empty com log")
        return
    for label, sel in [("all", rounds),
                       ("tentative (even)", [r for r in rounds if not r["commit"]]),
                       ("commit (odd)", [r for r in rounds if r["commit"]])]:
        if not sel:
            continue
        ts = sorted(r["t"] for r in sel)
        cs = [r["count"] for r in sel]
        q = lambda p: ts[min(len(ts) - 1, int(p * len(ts)))]
        print(f"{label:18s} n={len(sel):5d}  t(ms) mean={1e3*statistics.mean(ts):8.2f} "
              f"p50={1e3*q(0.5):8.2f} p95={1e3*q(0.95):8.2f} max={1e3*ts[-1]:8.2f}  "
              f"grads/round mean={statistics.mean(cs):.2f}")
    total = sum(r["t"] for r in rounds)
    print(f"total com-round wall time: {total:.2f}s over {len(rounds)} rounds")


if __name__ == "__main__":
    summarize(sys.argv[1])
