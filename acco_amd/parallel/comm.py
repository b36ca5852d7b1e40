"""Communication layer: RCCL over xGMI, bucketed; gloo fallback for CPU tests.

Implements the four-collective contract of the ACCO round (SURVEY.md §2.4
C1-C4) against a *bucket-major* flat-buffer layout:

    buffer = [nb buckets][world_size ranks][seg elements]

Each (bucket, rank) segment is contiguous, so every collective is a
contiguous in-place reduce-scatter / all-gather on one bucket. Rationale
(MI355X): collectives on ONE communicator execute in issue order — bucket
count does NOT add cross-collective link parallelism (RCCL spreads a
single collective over its channels/links by itself, controlled by
NCCL_MIN_NCHANNELS). What bucketing buys is the PIPELINE: the sharded
fused-AdamW for bucket j launches as soon as bucket j's reduce-scatter
lands and bucket j's all-gather launches right after its AdamW, so
optimizer compute and the remaining buckets' communication overlap, and
the first all-gather bytes are on the xGMI links long before the last
reduce-scatter finishes (the reference issues one monolithic pair with the
full optimizer step serialized between them, trainer_decoupled.py:88-112).
Per-bucket size also stays large enough (~34 MB at llama-1b world=8,
nb=8) to amortize per-collective launch cost on the 7×153 GB/s
point-to-point fabric; `comm_buckets` is the tuning knob.

The flat order of buffer[:N] is the model's parameter order, so
params ↔ buffer copies stay a single contiguous cast-copy; ownership of a
rank is nb strided-but-contiguous segments instead of one slice (invisible
to the algorithm: the fp32 optimizer shard is bucket-major contiguous).
"""

from __future__ import annotations

from dataclasses import dataclass


import torch
import torch.distributed as dist


@dataclass(frozen=True)
class ShardSpec:
    """Geometry of the bucket-major sharded flat buffer."""
    n: int          # live elements (model parameter count)
    world: int
    nb: int         # number of buckets
    seg: int        # per-(bucket, rank) segment elements

    @classmethod
    def build(cls, n: int, world: int, buckets: int = 8,
              align: int = 256) -> "ShardSpec":
        nb = max(1, int(buckets))
        per = (n + world * nb - 1) // (world * nb)
        seg = ((per + align - 1) // align) * align
        return cls(n=n, world=world, nb=nb, seg=seg)

    @property
    def bucket_elems(self) -> int:
        return self.world * self.seg

    @property
    def total(self) -> int:          # full buffer length B
        return self.nb * self.bucket_elems

    @property
    def owned(self) -> int:          # elements owned per rank (nb * seg)
        return self.nb * self.seg

    def bucket_view(self, buffer: torch.Tensor, j: int) -> torch.Tensor:
        o = j * self.bucket_elems
        return buffer[o:o + self.bucket_elems]

    def seg_view(self, buffer: torch.Tensor, j: int, rank: int) -> torch.Tensor:
        o = j * self.bucket_elems + rank * self.seg
        return buffer[o:o + self.seg]

    def owned_view(self, owned: torch.Tensor, j: int) -> torch.Tensor:
        return owned[j * self.seg:(j + 1) * self.seg]


class _NoopWork:
    def wait(self) -> None:
        pass


_NOOP = _NoopWork()


class CommBackend:
    """Collective operations for the ACCO/DDP rounds.

    backend "nccl" (= RCCL on ROCm) uses contiguous in-place
    reduce_scatter_tensor / all_gather_into_tensor; "gloo" (CPU tests)
    emulates reduce-scatter with an in-place all-reduce and all-gather with
    list-form all_gather, preserving semantics exactly.
    """

    def __init__(self, device: torch.device):
        self.device = device
        self.enabled = dist.is_available() and dist.is_initialized()
        self.world = dist.get_world_size() if self.enabled else 1
        self.backend = dist.get_backend() if self.enabled else "none"
        self._is_nccl = str(self.backend) == "nccl"

    # ---- init-time (C1): average random-init weights across ranks
    def all_reduce_avg(self, t: torch.Tensor) -> None:
        if self.world == 1:
            return
        if self._is_nccl:
            dist.all_reduce(t, op=dist.ReduceOp.AVG)
        else:
            # gloo has no AVG: SUM then scale (fp32 CPU, exact enough for init)
            dist.all_reduce(t, op=dist.ReduceOp.SUM)
            t.div_(self.world)

    # ---- grad-count all-reduce (C2)
    def all_reduce_sum_async(self, t: torch.Tensor):
        if self.world == 1:
            return _NOOP
        return dist.all_reduce(t, op=dist.ReduceOp.SUM, async_op=True)

    # ---- bucket reduce-scatter (C3)
    def reduce_scatter_bucket_async(self, buffer: torch.Tensor,
                                    spec: ShardSpec, j: int, rank: int):
        if self.world == 1:
            return _NOOP
        bucket = spec.bucket_view(buffer, j)
        if self._is_nccl:
            out = spec.seg_view(buffer, j, rank)
            return dist.reduce_scatter_tensor(out, bucket,
                                              op=dist.ReduceOp.SUM,
                                              async_op=True)
        # gloo: in-place all-reduce of the bucket — rank's segment then holds
        # the sum (other segments too; they are overwritten by all-gather).
        return dist.all_reduce(bucket, op=dist.ReduceOp.SUM, async_op=True)

    # ---- bucket all-gather (C4)
    def all_gather_bucket_async(self, buffer: torch.Tensor,
                                spec: ShardSpec, j: int, rank: int):
        if self.world == 1:
            return _NOOP
        bucket = spec.bucket_view(buffer, j)
        seg = spec.seg_view(buffer, j, rank)
        if self._is_nccl:
            return dist.all_gather_into_tensor(bucket, seg, async_op=True)
        views = [spec.seg_view(buffer, j, r) for r in range(self.world)]
        return dist.all_gather(views, seg.clone(), async_op=True)

    def barrier(self) -> None:
        if self.world > 1:
            dist.barrier()
