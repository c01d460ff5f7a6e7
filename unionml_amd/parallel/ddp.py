"""Single-node data parallelism: bucketed gradient all-reduce on RCCL.

One process per GPU; gradients all-reduce over xGMI through
``torch.distributed`` (backend ``"nccl"`` IS RCCL on ROCm). This is the
framework's own reducer, not ``torch.nn.parallel.DDP``:

- Parameter gradients are *views into flat per-bucket buffers*, so
  backward accumulates directly into communication-ready storage — no
  pack/unpack copies.
- Buckets fill in reverse parameter order (the order backward produces
  grads) and each bucket's all-reduce is issued asynchronously on the
  communication stream as soon as its last grad lands, overlapping the
  remaining backward compute.
- xGMI topology note: MI355X intra-node links are point-to-point
  (7 × ≈153 GB/s per GPU); ring collectives are per-link bound, so the
  default bucket is sized large (64 MiB) to amortize per-collective
  latency, and gradient dtype is kept as the compute dtype (bf16 grads
  all-reduce at half the bytes of fp32).

Works identically over ``gloo`` for CPU-only multi-process tests.

Known limitation (same class of issue torch DDP's
``find_unused_parameters`` exists for): every rank must run the same
model graph each step. If a parameter receives a gradient on some ranks
but not others (rank-divergent control flow), the per-bucket collectives
mismatch and the job can deadlock — shard DATA, not model structure.
"""

import os
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from unionml_amd._logging import logger

DEFAULT_BUCKET_MB = 64


def distributed_is_active() -> bool:
    return dist.is_available() and dist.is_initialized()


def get_rank() -> int:
    return dist.get_rank() if distributed_is_active() else 0


def get_world_size() -> int:
    return dist.get_world_size() if distributed_is_active() else 1


def shard(obj, rank: Optional[int] = None, world: Optional[int] = None):
    """Contiguous row-shard of a DataFrame/ndarray/tensor/list for this rank.

    Every rank gets ``ceil(n/world)`` or ``floor`` rows; uneven tails are
    balanced so ranks differ by at most one row.
    """
    rank = get_rank() if rank is None else rank
    world = get_world_size() if world is None else world
    if world <= 1:
        return obj
    n = len(obj)
    base, rem = divmod(n, world)
    start = rank * base + min(rank, rem)
    stop = start + base + (1 if rank < rem else 0)
    if hasattr(obj, "iloc"):  # DataFrame / Series
        return obj.iloc[start:stop]
    return obj[start:stop]


class _Bucket:
    def __init__(self, params: List[torch.nn.Parameter], dtype, device):
        self.params = params
        numel = sum(p.numel() for p in params)
        self.buffer = torch.zeros(numel, dtype=dtype, device=device)
        self.views: Dict[torch.nn.Parameter, torch.Tensor] = {}
        offset = 0
        for p in params:
            self.views[p] = self.buffer[offset : offset + p.numel()].view_as(p)
            offset += p.numel()
        self.pending = 0
        self.work = None

    def assign_grads(self):
        """Point each param's .grad at its bucket view so backward
        accumulates straight into the flat buffer."""
        for p in self.params:
            p.grad = self.views[p]
        self.pending = len(self.params)
        self.work = None


class GradientAllReducer:
    """Attach bucketed async all-reduce to a module's backward."""

    def __init__(
        self,
        module: torch.nn.Module,
        bucket_mb: float = DEFAULT_BUCKET_MB,
        process_group=None,
        sync_params: bool = True,
    ):
        self.module = module
        self.group = process_group
        self.world = get_world_size()
        params = [p for p in module.parameters() if p.requires_grad]
        device = params[0].device if params else torch.device("cpu")

        # ranks must start from identical weights: broadcast rank 0's
        # parameters and buffers (BN running stats) once at attach time
        if sync_params and self.world > 1 and distributed_is_active():
            with torch.no_grad():
                for t in list(module.parameters()) + list(module.buffers()):
                    dist.broadcast(t.data, src=0, group=self.group)

        # reverse registration order ~ the order backward produces grads
        self.buckets: List[_Bucket] = []
        bucket_bytes = int(bucket_mb * 1024 * 1024)
        cur: List[torch.nn.Parameter] = []
        cur_bytes = 0
        for p in reversed(params):
            psize = p.numel() * p.element_size()
            if cur and cur_bytes + psize > bucket_bytes:
                self.buckets.append(_Bucket(cur, cur[0].dtype, device))
                cur, cur_bytes = [], 0
            if cur and cur[0].dtype != p.dtype:
                self.buckets.append(_Bucket(cur, cur[0].dtype, device))
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += psize
        if cur:
            self.buckets.append(_Bucket(cur, cur[0].dtype, device))

        self.param_bucket: Dict[torch.nn.Parameter, _Bucket] = {}
        for b in self.buckets:
            for p in b.params:
                self.param_bucket[p] = b

        self._hooks = [
            p.register_post_accumulate_grad_hook(self._on_grad_ready) for p in params
        ]
        self.zero_grad()  # installs the grad views

    def _on_grad_ready(self, param: torch.nn.Parameter):
        bucket = self.param_bucket[param]
        bucket.pending -= 1
        if bucket.pending == 0 and self.world > 1 and distributed_is_active():
            bucket.work = dist.all_reduce(
                bucket.buffer, op=dist.ReduceOp.SUM, group=self.group, async_op=True
            )

    def zero_grad(self):
        for b in self.buckets:
            b.buffer.zero_()
            b.assign_grads()

    def finalize(self):
        """Wait for all in-flight reductions and average. Call between
        ``loss.backward()`` and ``optimizer.step()``."""
        if self.world <= 1 or not distributed_is_active():
            return
        inv = 1.0 / self.world
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
                b.work = None
            elif b.pending != 0 and b.pending != len(b.params):
                # partial bucket (e.g. frozen params this step): reduce now
                dist.all_reduce(b.buffer, op=dist.ReduceOp.SUM, group=self.group)
            b.buffer.mul_(inv)

    def detach(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []


def maybe_wrap(module: torch.nn.Module, bucket_mb: float = DEFAULT_BUCKET_MB):
    """Attach a :class:`GradientAllReducer` when running under an active
    process group; no-op (returns ``None``) in single-process runs.

    Usage inside a trainer body::

        reducer = maybe_wrap(model)
        for batch in data:
            loss = ...
            loss.backward()
            if reducer: reducer.finalize()
            optimizer.step()
            (reducer.zero_grad() if reducer else optimizer.zero_grad())
    """
    if get_world_size() <= 1:
        return None
    return GradientAllReducer(module, bucket_mb=bucket_mb)
