"""Native data-parallel engine: bucketed gradient all-reduce over RCCL.

This replaces `torch.nn.parallel.DistributedDataParallel` (the reference
family's only distribution mechanism) with our own DP
engine designed for the MI355X node topology:

- one process per GPU; `torch.distributed` backend "nccl" IS RCCL on
  ROCm, riding the fully-connected 7x153 GB/s xGMI fabric
- gradients are packed into a FEW LARGE flat buckets (default 25 MB —
  the whole generator fits in ~2): total gradient volume is tiny
  (<=40 MB fp32/step) so ring-all-reduce LATENCY, not bandwidth,
  dominates on xGMI — few launches beat many
- the bucket views ARE the gradient storage: `zero_grad()` binds each
  `p.grad` to its slice of the flat bucket, so backward accumulates
  straight into the bucket and `finish()` has nothing to copy (the
  round-1 version copied grad->bucket and averaged-bucket->grad — two
  full-gradient passes per step on the latency-critical path)
- each bucket's all-reduce is launched asynchronously the moment its
  last gradient is accumulated (post-accumulate-grad hooks), so comm
  runs on RCCL's comm stream concurrently with the REST of backward
- `finish()` waits the handles before optimizer.step() and records a
  per-bucket (ready -> done) timeline so multi-GPU runs emit an
  overlap artifact (how much comm was already complete when backward
  ended) instead of just a throughput number

CPU CI runs the same engine over the gloo backend (world_size>1 multi-
process tests), so the distributed path is covered without
a GPU.
"""

from __future__ import annotations

import time

import torch
import torch.distributed as dist


def init_distributed(backend: str | None = None) -> tuple[int, int]:
    """Initialise from torchrun env. Returns (rank, world_size)."""
    import os

    if dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world <= 1:
        return 0, 1
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    dist.init_process_group(backend=backend)
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    return dist.get_rank(), dist.get_world_size()


class GradBucket:
    def __init__(self, params: list[torch.nn.Parameter], device, dtype):
        self.params = params
        self.numel = sum(p.numel() for p in params)
        self.flat = torch.zeros(self.numel, device=device, dtype=dtype)
        self.views = []
        off = 0
        for p in params:
            self.views.append(self.flat.narrow(0, off, p.numel())
                              .view_as(p))
            off += p.numel()
        self.ready = 0
        self.work = None
        self.t_ready = 0.0
        self.t_done = 0.0

    def reset(self):
        self.ready = 0
        self.work = None


class BucketAllReduceDP:
    """Wraps a module; averages gradients across ranks, overlapped.

    Usage:
        engine = BucketAllReduceDP(model, bucket_cap_mb=25)
        engine.zero_grad(optimizer)   # zero buckets + bind p.grad -> views
        loss.backward()               # hooks launch async all-reduces
        engine.finish()               # wait handles (grads already in place)
        optimizer.step()
    """

    def __init__(self, module: torch.nn.Module, bucket_cap_mb: float = 25.0,
                 process_group=None):
        self.module = module
        self.group = process_group
        self.world = dist.get_world_size(process_group) \
            if dist.is_initialized() else 1
        self.enabled = self.world > 1
        self._hooks = []
        self.buckets: list[GradBucket] = []
        self._param_bucket: dict[int, tuple[GradBucket, int]] = {}
        self.last_stats: dict = {}
        if not self.enabled:
            return

        self.broadcast_parameters()

        # Build buckets in REVERSE parameter order — gradients become
        # ready roughly output-to-input during backward, so reverse order
        # lets the first bucket fill (and its all-reduce launch) earliest.
        params = [p for p in module.parameters() if p.requires_grad]
        cap = int(bucket_cap_mb * 1024 * 1024)
        cur: list[torch.nn.Parameter] = []
        cur_bytes = 0
        groups: list[list[torch.nn.Parameter]] = []
        for p in reversed(params):
            sz = p.numel() * p.element_size()
            # p.grad will BE a view of the flat bucket, so a bucket must be
            # dtype-homogeneous
            if cur and (cur_bytes + sz > cap or p.dtype != cur[0].dtype):
                groups.append(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += sz
        if cur:
            groups.append(cur)
        device = params[0].device
        for g in groups:
            # grad storage shares the param dtype so p.grad can BE the view
            b = GradBucket(g, device, g[0].dtype)
            self.buckets.append(b)
            for i, p in enumerate(g):
                self._param_bucket[id(p)] = (b, i)
        self._bind_grads()

        for p in params:
            h = p.register_post_accumulate_grad_hook(self._on_grad)
            self._hooks.append(h)

    def broadcast_parameters(self):
        """rank0 -> all at construction (DDP contract)."""
        if not self.enabled:
            return
        for t in self.module.state_dict().values():
            if isinstance(t, torch.Tensor) and t.numel() > 0:
                dist.broadcast(t.data, src=0, group=self.group)

    def _bind_grads(self):
        for b in self.buckets:
            for p, v in zip(b.params, b.views):
                p.grad = v

    def zero_grad(self, optimizer=None):
        """Zero gradients for the wrapped module's next step.

        Enabled: zero the flat buckets and (re)bind p.grad to the bucket
        views so backward accumulates in place — pointer-stable across
        steps (FusedAdam's cached device tables and hipGraph capture both
        rely on that).  Disabled (world 1): plain set_to_none semantics
        via the optimizer if given, else per-param.
        """
        if not self.enabled:
            if optimizer is not None:
                optimizer.zero_grad(set_to_none=True)
            else:
                for p in self.module.parameters():
                    p.grad = None
            return
        for b in self.buckets:
            b.flat.zero_()
            b.reset()
        self._bind_grads()

    def _on_grad(self, p: torch.nn.Parameter):
        b, i = self._param_bucket[id(p)]
        if p.grad is not None and p.grad.data_ptr() != b.views[i].data_ptr():
            # a backward ran without zero_grad() (autograd allocated fresh
            # storage) — fold it into the bucket and rebind
            b.views[i].add_(p.grad.detach().to(b.flat.dtype))
            p.grad = b.views[i]
        b.ready += 1
        if b.ready == len(b.params):
            b.flat.div_(self.world)
            b.t_ready = time.perf_counter()
            b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM,
                                     group=self.group, async_op=True)

    def finish(self):
        """Wait outstanding all-reduces (averaged grads are already in
        p.grad — the views are the storage). Records overlap stats."""
        if not self.enabled:
            return
        t_bwd_end = time.perf_counter()
        launched = [b for b in self.buckets if b.work is not None]
        pre_done = 0
        for b in self.buckets:
            if b.ready not in (0, len(b.params)):
                raise RuntimeError(
                    f"DP bucket filled {b.ready}/{len(b.params)} times: a "
                    "backward touched this module's params twice without "
                    "finish(), or a cross-model backward leaked into it "
                    "(freeze the other model's params — see "
                    "Trainer._forward_g)")
            if b.work is not None:
                if b.work.is_completed():
                    pre_done += 1
                b.work.wait()
                b.t_done = time.perf_counter()
        t_end = time.perf_counter()
        if launched:
            self.last_stats = {
                "buckets": len(launched),
                "bytes": sum(b.numel * b.flat.element_size()
                             for b in launched),
                "pre_completed": pre_done,
                "finish_wait_ms": round((t_end - t_bwd_end) * 1e3, 3),
                "bucket_ms": [round((b.t_done - b.t_ready) * 1e3, 3)
                              for b in launched],
            }
        for b in self.buckets:
            b.reset()

    def comm_bytes_per_step(self) -> int:
        return sum(b.numel * b.flat.element_size() for b in self.buckets)

    def remove(self):
        for h in self._hooks:
            h.remove()
        self._hooks.clear()
