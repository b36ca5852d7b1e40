"""Scaling-table harness: runs bench.py at N=1..8 GPUs for acco and ddp and
writes benchmarks/scaling.json + a markdown table (the
ACCO-vs-DDP wall-clock speedup and tokens/s/node scaling curve are the
headline numbers this repo must self-measure).

Run ON a GPU node:
    python benchmarks/run_scaling.py --gpus 1 2 4 8 --steps 10 --warmup 3
"""

from __future__ import annotations

import argparse
import json
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def run_bench(n: int, steps: int, warmup: int, method: str, model: str,
              port: int = 29581) -> dict:
    if n == 1:
        cmd = [sys.executable, "bench.py"]
    else:
        cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
               f"--nproc-per-node={n}", "--master-addr", "127.0.0.1",
               "--master-port", str(port), "bench.py"]
    cmd += ["--gpus", str(n), "--steps", str(steps), "--warmup", str(warmup),
            "--method", method, "--model", model]
    out = subprocess.run(cmd, cwd=HERE, capture_output=True, text=True,
                         timeout=1800)
    for line in reversed(out.stdout.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise RuntimeError(f"no JSON from bench (n={n} {method}): "
                       f"{out.stdout[-2000:]}\n{out.stderr[-2000:]}")


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, nargs="+", default=[1, 2, 4, 8])
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--model", default="llama-1b")
    ap.add_argument("--methods", nargs="+", default=["acco", "ddp"])
    ap.add_argument("--out", default=os.path.join(HERE, "benchmarks",
                                                  "scaling.json"))
    args = ap.parse_args()

    results = []
    for n in args.gpus:
        for method in args.methods:
            r = run_bench(n, args.steps, args.warmup, method, args.model)
            results.append(r)
            print(json.dumps(r))

    with open(args.out, "w") as f:
        json.dump(results, f, indent=2)

    # markdown table
    lines = ["| GPUs | method | tokens/s | ms/step | speedup acco/ddp |",
             "|---|---|---|---|---|"]
    by = {(r["n_gpus"], r["config"]["method"]): r for r in results}
    for n in args.gpus:
        for method in args.methods:
            r = by.get((n, method))
            if r is None:
                continue
            sp = ""
            if method == "acco" and (n, "ddp") in by:
                sp = f'{r["value"] / by[(n, "ddp")]["value"]:.3f}'
            lines.append(f'| {n} | {method} | {r["value"]:.0f} | '
                         f'{r["ms_per_step"]:.1f} | {sp} |')
    table = "\n".join(lines)
    print(table)
    with open(os.path.join(HERE, "benchmarks", "scaling.md"), "w") as f:
        f.write(table + "\n")


if __name__ == "__main__":
    main()
