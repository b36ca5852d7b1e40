"""Observability: scalar logging, results CSV, stdout progress.

Covers the reference's three channels (utils/logs_utils.py): TensorBoard
scalars (:187-224) → a JSONL ScalarLogger (TensorBoard is not installed in
this environment; the schema keeps the same scalar names), results.csv
merge-on-append (:71-138), and throttled stdout progress (:155-183).
"""

from __future__ import annotations

import csv
import json
import logging
import os
import time
import uuid
from typing import Any, Dict, Optional


def get_logger(name: str = "acco_amd") -> logging.Logger:
    log = logging.getLogger(name)
    if not log.handlers:
        h = logging.StreamHandler()
        h.setFormatter(logging.Formatter("[%(asctime)s %(name)s %(levelname)s] %(message)s"))
        log.addHandler(h)
        log.setLevel(logging.INFO)
    return log


def create_id_run() -> str:
    """A unique run id (the reference uses SLURM_JOBID, trainer_base.py:139;
    we fall back to a uuid when not under SLURM)."""
    return os.environ.get("SLURM_JOBID") or uuid.uuid4().hex[:12]


class ScalarLogger:
    """JSONL scalar stream, one file per run: the TensorBoard-channel
    equivalent (reference utils/logs_utils.py:187-224)."""

    def __init__(self, log_dir: str, run_name: str, id_run: str, rank: int = 0,
                 tensorboard: bool = True):
        self.rank = rank
        self.path: Optional[str] = None
        self._f = None
        self._tb = None
        if rank == 0:
            d = os.path.join(log_dir, run_name)
            os.makedirs(d, exist_ok=True)
            self.path = os.path.join(d, f"{id_run}.jsonl")
            self._f = open(self.path, "a", buffering=1)
            if tensorboard:
                from acco_amd.utils.tb_writer import EventFileWriter
                self._tb = EventFileWriter(os.path.join(d, "tb", id_run))

    def add_scalar(self, tag: str, value: float, step: int) -> None:
        if self._f is None:
            return
        self._f.write(json.dumps({"tag": tag, "value": float(value),
                                  "step": int(step), "ts": time.time()}) + "\n")
        if self._tb is not None:
            self._tb.add_scalar(tag, value, step)

    def add_histogram(self, tag: str, values, step: int) -> None:
        """Distribution logging (grad/param histograms) — TB-only channel."""
        if self._tb is not None:
            self._tb.add_histogram(tag, values, step)

    def log_training(self, opt_step: int, n_grads: int, rank: int, loss: float,
                     eval_loss: Optional[float], t_beg: float) -> None:
        """Scalar-name schema mirrors reference log_to_tensorboard
        (loss_t / loss_step / loss_samples keyed by rank)."""
        t = time.time() - t_beg
        self.add_scalar(f"loss_t/{rank}", loss, int(t))
        self.add_scalar(f"loss_step/{rank}", loss, opt_step)
        self.add_scalar(f"loss_samples/{rank}", loss, n_grads)
        if eval_loss is not None:
            self.add_scalar(f"eval_loss_step/{rank}", eval_loss, opt_step)

    def close(self) -> None:
        if self._f is not None:
            self._f.close()
            self._f = None
        if self._tb is not None:
            self._tb.close()
            self._tb = None


def create_dict_result(dict_args: Dict[str, Any], world_size: int, n_nodes: int,
                       device_name: str, total_time: float, id_run: str,
                       loss_final: float) -> Dict[str, Any]:
    """One result row per run (reference utils/logs_utils.py:43-68 schema)."""
    row = dict(dict_args)
    row.update({
        "Id_run": id_run,
        "N_workers": world_size,
        "n_nodes": n_nodes,
        "cuda_device": device_name,
        "Tot_time": total_time,
        "Loss_final": float(loss_final),
    })
    return row


def save_result(path: str, row: Dict[str, Any]) -> None:
    """Append a row to results.csv, merging schemas across runs
    (reference utils/logs_utils.py:71-138 behaviour: schema union)."""
    rows = []
    fields: list = []
    if os.path.exists(path):
        with open(path, newline="") as f:
            reader = csv.DictReader(f)
            fields = list(reader.fieldnames or [])
            rows = list(reader)
    for k in row:
        if k not in fields:
            fields.append(k)
    rows.append({k: row.get(k, "") for k in fields})
    with open(path, "w", newline="") as f:
        writer = csv.DictWriter(f, fieldnames=fields)
        writer.writeheader()
        for r in rows:
            writer.writerow({k: r.get(k, "") for k in fields})


def print_training_evolution(log: logging.Logger, count_grad_tot: int, count_com: int,
                             delta_step: int, rank: int, t_beg: float,
                             t_last: float, loss: float, epoch: int):
    """Throttled stdout progress (reference utils/logs_utils.py:155-183):
    one line whenever count_grad_tot crosses the next `delta_step` multiple."""
    if count_grad_tot // delta_step > epoch:
        now = time.time()
        log.info(
            f"grads={count_grad_tot} coms={count_com} loss={loss:.4f} "
            f"dt={now - t_last:.2f}s total={now - t_beg:.1f}s"
        )
        return count_grad_tot // delta_step, now
    return epoch, t_last
