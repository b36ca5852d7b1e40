"""Flagship benchmark: Llama-style 1B ACCO training step on MI355X.

Driver contract:
    python bench.py --gpus N --steps K --warmup W
(N>1 is launched by the driver via torch.distributed.run, one rank per GPU
over RCCL; RANK/LOCAL_RANK/WORLD_SIZE/MASTER_* read from the env.)

A "step" is one ACCO com round (count all-reduce + bucketed grad
reduce-scatter + sharded fused AdamW + param all-gather, overlapped with
the concurrent micro-batch forward/backward) — two rounds = one tentative +
one true optimizer step, the algorithm's steady-state cadence. For
--method ddp a step is one synchronous optimizer step. The headline metric
is whole-job tokens/s (micro-batches executed in the timed window × batch ×
seq, summed over ranks, divided by the max-over-ranks wall time).

Timing bracket: torch.cuda.synchronize() + a gloo control-plane barrier on
both sides of EXACTLY K steps (the gloo group keeps the bracket off the
RCCL communicator that the com thread is using concurrently).
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch
import torch.distributed as dist

MODELS = {
    # flagship (BASELINE.json config 3): Llama-3.2-1B-shaped trunk,
    # OpenWebText-sized vocab (see acco_amd/config/model/llama-1b.yaml)
    "llama-1b": dict(hidden_size=2048, num_layers=16, num_heads=32,
                     num_kv_heads=8, intermediate_size=8192,
                     vocab_size=50304, max_position_embeddings=4096,
                     tie_word_embeddings=True),
    "llama-8b": dict(hidden_size=4096, num_layers=32, num_heads=32,
                     num_kv_heads=8, intermediate_size=14336,
                     vocab_size=128256, max_position_embeddings=8192,
                     tie_word_embeddings=False),
    # small model for CPU/debug runs
    "tiny": dict(hidden_size=256, num_layers=2, num_heads=4, num_kv_heads=2,
                 intermediate_size=512, vocab_size=1024,
                 max_position_embeddings=2048, tie_word_embeddings=True),
}

# GPT-Neo family entries (BASELINE.json configs 1-2: GPT-2-small-shaped)
NEO_MODELS = {
    "gptneo-125m": dict(hidden_size=768, num_layers=12, num_heads=12,
                        vocab_size=50304, max_position_embeddings=2048,
                        window_size=256),
}


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--method", choices=["acco", "ddp", "dpu"], default="acco")
    p.add_argument("--model", default="llama-1b")
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--seq", type=int, default=1024)
    p.add_argument("--buckets", type=int, default=8)
    p.add_argument("--lr", type=float, default=6e-4)
    p.add_argument("--ckpt", action="store_true",
                   help="per-layer activation recompute (8B batch headroom)")
    return p.parse_args()


def make_batch_pool(n_pool, batch, seq, vocab, device, seed):
    g = torch.Generator().manual_seed(seed)
    pool = [torch.randint(0, vocab, (batch, seq), generator=g) for _ in range(n_pool)]
    if device.type == "cuda":
        pool = [t.pin_memory() for t in pool]
    return pool


def main():
    args = parse_args()
    # dmabuf IPC (see engine/bootstrap.py) — required for multi-rank RCCL
    os.environ.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env_world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    world = max(env_world, 1)
    if args.gpus != world:
        # refuse to emit a record labeled n_gpus=K while actually running
        # on WORLD_SIZE=W ranks (the driver launches one rank per GPU)
        print(f"--gpus {args.gpus} != WORLD_SIZE {world}; launch with "
              f"torch.distributed.run --nproc-per-node {args.gpus}",
              file=sys.stderr)
        sys.exit(2)

    cuda = torch.cuda.is_available()
    if cuda:
        dev_idx = local_rank % max(1, torch.cuda.device_count())
        torch.cuda.set_device(dev_idx)
        device = torch.device("cuda", dev_idx)
        backend = "nccl"
    else:
        device = torch.device("cpu")
        backend = "gloo"

    if world > 1 or "MASTER_ADDR" in os.environ:
        if not dist.is_initialized():
            os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
            os.environ.setdefault("MASTER_PORT", "29531")
            dist.init_process_group(backend, rank=rank, world_size=world)
        ctrl = dist.new_group(backend="gloo")
    else:
        ctrl = None

    from acco_amd.engine import arena
    from acco_amd.engine.acco import AccoEngine
    from acco_amd.engine.scheduler import LRSchedule
    from acco_amd.engine.sharded_adamw import ShardedAdamW
    from acco_amd.models import (GPTNeoConfig, GPTNeoForCausalLM,
                                 LlamaConfig, LlamaForCausalLM)
    from acco_amd.parallel.comm import CommBackend, ShardSpec
    from acco_amd.parallel.ddp import NativeZeroDDP

    model_name = args.model if cuda else "tiny"
    torch.manual_seed(42)
    if model_name in NEO_MODELS:
        mcfg = GPTNeoConfig(**NEO_MODELS[model_name])
        model = GPTNeoForCausalLM(mcfg)
    else:
        mcfg = LlamaConfig(**MODELS[model_name])
        model = LlamaForCausalLM(mcfg)
    dtype = torch.bfloat16 if cuda else torch.float32

    n_live = arena.live_numel(model)
    spec = ShardSpec.build(n_live, world, buckets=args.buckets)
    params = arena.flatten_params(model, dtype, device, pad_to=spec.total)
    grads = arena.attach_grad_arena(model, dtype, device, pad_to=spec.total)
    comm = CommBackend(device)
    comm.all_reduce_avg(params)

    # fused projections run in every mode (DDP readiness tracking is
    # element-coverage based) so ACCO vs DDP compares identical compute
    from acco_amd.models.fuse import install_fused_projections
    n_fused = install_fused_projections(model, params, grads)
    if rank == 0:
        print(f"# fused projection groups: {n_fused}", file=sys.stderr)
    if args.ckpt:
        model.model.gradient_checkpointing = True

    opt = ShardedAdamW(spec, rank, device, lr=args.lr, betas=(0.9, 0.95),
                       eps=1e-8, weight_decay=0.1)
    opt.init_master_from_buffer(params)
    sched = LRSchedule(args.lr, 1000, 50_000, "cosine")

    pool = make_batch_pool(8, args.batch, args.seq, mcfg.vocab_size, device,
                           seed=1234 + rank)
    pool_idx = [0]

    def next_batch():
        t = pool[pool_idx[0] % len(pool)]
        pool_idx[0] += 1
        return {"input_ids": t.to(device, non_blocking=True)}

    def forward_backward(inputs):
        ids = inputs["input_ids"]
        loss, _ = model(ids, labels=ids)
        loss.backward()
        return loss.detach()

    def ctrl_barrier():
        if ctrl is not None:
            dist.barrier(group=ctrl)

    def device_sync():
        if cuda:
            torch.cuda.synchronize()

    warmup = max(args.warmup, 1)
    steps = args.steps
    state = {"t0": None, "t1": None, "mb0": 0, "mb1": 0}

    if args.method in ("acco", "dpu"):
        eng = AccoEngine(params_arena=params, grads_arena=grads,
                         n_live=n_live, spec=spec, comm=comm, rank=rank,
                         device=device, opt=opt, sched=sched,
                         forward_backward=forward_backward,
                         next_batch=next_batch, n_grad_accumulation=1)

        def boundary(round_idx):
            if round_idx == warmup:
                device_sync(); ctrl_barrier()
                state["t0"] = time.time(); state["mb0"] = eng.micro_steps
            elif round_idx == warmup + steps:
                device_sync(); ctrl_barrier()
                state["t1"] = time.time(); state["mb1"] = eng.micro_steps

        eng.on_round_boundary = boundary
        eng.enable_arena_swap(model, grads)
        if args.method == "acco":
            eng.train_acco(nb_grad_tot=1 << 60, n_warmup_steps=0,
                           max_rounds=warmup + steps)
        else:
            eng.train_dpu(nb_grad_tot=1 << 60, n_warmup_steps=0,
                          max_rounds=warmup + steps)
        micro = state["mb1"] - state["mb0"]
    else:
        ddp = NativeZeroDDP(model, params, grads, n_live, spec, comm, rank, opt)
        micro_count = 0

        def ddp_step():
            nonlocal micro_count
            ddp.begin_sync_microbatch()
            ids = next_batch()["input_ids"]
            loss, _ = model(ids, labels=ids)
            loss.backward()
            micro_count += 1
            ddp.finish_step(grad_scale=1.0 / world, lr=sched.lr())
            ddp.zero_grad()
            sched.advance(world)

        for _ in range(warmup):
            ddp_step()
        device_sync(); ctrl_barrier()
        state["t0"] = time.time(); state["mb0"] = micro_count
        for _ in range(steps):
            ddp_step()
        device_sync(); ctrl_barrier()
        state["t1"] = time.time(); state["mb1"] = micro_count
        micro = state["mb1"] - state["mb0"]

    elapsed = state["t1"] - state["t0"]
    tokens_local = micro * args.batch * args.seq

    if ctrl is not None:
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX, group=ctrl)
        elapsed = float(t)
        tk = torch.tensor([tokens_local], dtype=torch.float64)
        dist.all_reduce(tk, op=dist.ReduceOp.SUM, group=ctrl)
        tokens_total = float(tk)
    else:
        tokens_total = float(tokens_local)

    if rank == 0:
        value = tokens_total / elapsed
        print(json.dumps({
            "metric": "tokens/s",
            "value": value,
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": steps,
            "warmup": warmup,
            "ms_per_step": elapsed / steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "bf16" if cuda else "fp32",
            "data": "synthetic",
            "config": {
                "model": model_name,
                "method": args.method,
                "global_batch": args.batch * world,
                "seq_len": args.seq,
                "parallelism": (f"{args.method}-dp{world}"
                                if args.method != "ddp" else f"ddp{world}"),
                "comm_buckets": spec.nb,
                "params": n_live,
                "ckpt": bool(args.ckpt),
                "peak_mem_gb": (round(torch.cuda.max_memory_allocated()
                                      / 2**30, 1) if cuda else None),
            },
        }))

    if dist.is_initialized():
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
