"""Data-parallel gradient synchronization: bucketed all-reduce over RCCL/xGMI
overlapped with backward (SURVEY.md §2.2 DP row).

Instead of wrapping the model in torch DDP, gradients live in the optimizer's
flat arenas (optim.py) and are reduced as contiguous arena slices: parameters
are bucketed by arena offset, each parameter's post-accumulate-grad hook
counts its bucket down, and a full bucket launches an async all-reduce that
overlaps with the rest of backward.  Bucket size is tuned for the xGMI mesh
(7 p2p links x ~153 GB/s per GPU -> large buckets amortize per-link ring
latency; default 128 MB).

Also works without arenas (plain optimizers): per-parameter async all-reduce.
"""

import contextlib
from typing import List, Optional

import torch
import torch.distributed as dist


class Bucket:
    def __init__(self, flat: Optional[torch.Tensor], lo: int, hi: int, params: List[torch.nn.Parameter]):
        self.flat = flat
        self.lo = lo
        self.hi = hi
        self.params = set(params)
        self.pending = len(params)
        self.handle = None

    def reset(self):
        self.pending = len(self.params)
        self.handle = None


class GradReducer:
    """Bucketed async gradient all-reduce driven by post-accumulate hooks."""

    def __init__(self, optimizer, model: torch.nn.Module, bucket_size_mb: int = 128,
                 process_group=None, average: bool = False, zero: bool = False):
        """``average=False`` assumes the optimizer fuses the 1/N scaling
        (FusedAdamW.grad_scale); ``average=True`` divides after reduction.
        ``zero=True``: instead of bucketed all-reduce, finalize() reduce-
        scatters each grad arena into this rank's shard (the sharded optimizer
        consumes only the shard)."""
        self.enabled = dist.is_initialized() and dist.get_world_size(process_group) > 1
        self.group = process_group
        self.average = average
        self.zero = zero and self.enabled
        self.optimizer = optimizer
        self.buckets: List[Bucket] = []
        self._param_bucket = {}
        self._sync = True
        self._hooks = []
        if not self.enabled or self.zero:
            return

        bucket_numel_budget = bucket_size_mb * 1024 * 1024 // 2  # bf16 elems

        if hasattr(optimizer, "grad_arenas"):
            for flat_g, entries in optimizer.grad_arenas():
                # contiguous runs of the arena, capped by the bucket budget
                cur_params, lo = [], None
                for p, off, n in entries:
                    if lo is None:
                        lo = off
                    cur_params.append(p)
                    if off + n - lo >= bucket_numel_budget:
                        self._add_bucket(flat_g, lo, off + n, cur_params)
                        cur_params, lo = [], None
                if cur_params:
                    last_p, last_off, last_n = entries[-1]
                    self._add_bucket(flat_g, lo, last_off + last_n, cur_params)
        else:
            for p in model.parameters():
                if p.requires_grad:
                    self._add_bucket(None, 0, 0, [p])

        for bucket in self.buckets:
            for p in bucket.params:
                self._param_bucket[p] = bucket
                self._hooks.append(p.register_post_accumulate_grad_hook(self._on_grad))

    def _add_bucket(self, flat, lo, hi, params):
        self.buckets.append(Bucket(flat, lo, hi, list(params)))

    def _on_grad(self, param):
        if not self._sync or not self.enabled:
            return
        bucket = self._param_bucket[param]
        bucket.pending -= 1
        if bucket.pending == 0:
            if bucket.flat is not None:
                tensor = bucket.flat[bucket.lo : bucket.hi]
            else:
                tensor = next(iter(bucket.params)).grad
            bucket.handle = dist.all_reduce(tensor, dist.ReduceOp.SUM, group=self.group, async_op=True)

    def finalize(self):
        """Wait for in-flight reductions; call after backward, before step."""
        if not self.enabled:
            return
        world = dist.get_world_size(self.group)
        if self.zero:
            # ZeRO: one reduce-scatter per grad arena; the sharded optimizer
            # reads only [lo, lo+shard)
            for group_arenas in self.optimizer._arenas:
                for arena in group_arenas:
                    flat_g = arena["flat_g"]
                    lo, shard = arena["lo"], arena["shard"]
                    if dist.get_backend(self.group) == "gloo":
                        dist.all_reduce(flat_g, dist.ReduceOp.SUM, group=self.group)
                    else:
                        out = torch.empty_like(flat_g[lo : lo + shard])
                        dist.reduce_scatter_tensor(out, flat_g, dist.ReduceOp.SUM, group=self.group)
                        flat_g[lo : lo + shard].copy_(out)
                    if self.average:
                        flat_g[lo : lo + shard].div_(world)
            return
        for bucket in self.buckets:
            if bucket.handle is not None:
                bucket.handle.wait()
            elif self._sync and bucket.pending > 0 and bucket.pending < len(bucket.params):
                raise RuntimeError("GradReducer: bucket partially fired — hook/bucket mismatch")
            elif self._sync and bucket.pending == len(bucket.params):
                # params that saw no grad this step (e.g. unused head): reduce anyway
                if bucket.flat is not None:
                    dist.all_reduce(bucket.flat[bucket.lo : bucket.hi], dist.ReduceOp.SUM, group=self.group)
                else:
                    p = next(iter(bucket.params))
                    if p.grad is not None:
                        dist.all_reduce(p.grad, dist.ReduceOp.SUM, group=self.group)
            if self.average:
                if bucket.flat is not None:
                    bucket.flat[bucket.lo : bucket.hi].div_(world)
                else:
                    p = next(iter(bucket.params))
                    if p.grad is not None:
                        p.grad.div_(world)
            bucket.reset()

    @contextlib.contextmanager
    def no_sync(self):
        """Accumulate grads locally without reduction (grad accumulation;
        reference accelerate_base_trainer.py:502-516)."""
        prev = self._sync
        self._sync = False
        try:
            yield
        finally:
            self._sync = prev

    def broadcast_parameters(self, model: torch.nn.Module, src: int = None):
        """Ensure identical initial weights across ranks (random-init runs).
        ``src`` is a GLOBAL rank; defaults to the group's first member (the
        group may not contain global rank 0 under model parallelism)."""
        if not self.enabled:
            return
        if src is None:
            if self.group is not None:
                src = dist.get_process_group_ranks(self.group)[0]
            else:
                src = 0
        for p in model.state_dict().values():
            if isinstance(p, torch.Tensor):
                dist.broadcast(p, src, group=self.group)
