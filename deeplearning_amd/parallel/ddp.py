"""BucketedDataParallel — DDP-semantics gradient data-parallelism designed for
RCCL over xGMI (MI355X node: 7 point-to-point links/GPU, ~153 GB/s each).

Design (SURVEY.md §2.3 design note): gradients accumulate into param.grad as
usual; per-param post-accumulate hooks mark bucket members ready; a full bucket
is flattened into a preallocated flat buffer and all-reduced asynchronously
(RCCL schedules on its own stream -> overlaps the remaining backward). Buckets
are filled in reverse parameter order (approximate backward order). Bucket size
defaults to 64 MiB — sized by xGMI arithmetic, not yet swept on an 8-GPU node
(single-GPU rounds): a 64 MiB ring step is far above RCCL's latency floor and
the whole-model all-reduce (~1-4 ms for resnet50/vit_b16 at ring-8 link speed)
hides under a 23-46 ms backward with >=2 buckets of overlap. `bench.py
--bucket-mb` sweeps 16/32/64/128 when a multi-GPU node is available.

Differences from torch DDP kept deliberately: no graph rebuilding, no
find_unused_parameters machinery (CV models here are static), SUM+divide (works
on gloo for CPU tests) instead of premultiplied AVG.
"""
from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn

from ..core.dist import get_world_size, is_dist


class _Bucket:
    __slots__ = ("params", "flat", "ready", "work", "views")

    def __init__(self, params, device, dtype):
        self.params = params
        numel = sum(p.numel() for p in params)
        self.flat = torch.zeros(numel, device=device, dtype=dtype)
        self.views = []
        off = 0
        for p in params:
            self.views.append(self.flat.narrow(0, off, p.numel()).view_as(p))
            off += p.numel()
        self.ready = 0
        self.work = None


class BucketedDataParallel(nn.Module):
    def __init__(self, module: nn.Module, bucket_cap_mb: float = 64.0,
                 broadcast_params: bool = True, process_group=None,
                 grad_dtype: torch.dtype | None = None):
        super().__init__()
        self.module = module
        self.pg = process_group
        self.world_size = get_world_size() if process_group is None else dist.get_world_size(process_group)
        self.enabled = is_dist() and self.world_size > 1
        self._require_sync = True
        self._hooks = []

        if self.enabled and broadcast_params:
            with torch.no_grad():
                for t in list(module.parameters()) + list(module.buffers()):
                    # expert-parallel params are rank-LOCAL (each rank owns
                    # its expert shard): neither broadcast nor all-reduce
                    if getattr(t, "expert", False):
                        continue
                    dist.broadcast(t, src=0, group=self.pg)

        self.buckets: list[_Bucket] = []
        self._param_bucket: dict[int, _Bucket] = {}
        if self.enabled:
            self._build_buckets(bucket_cap_mb, grad_dtype)

    def _build_buckets(self, cap_mb: float, grad_dtype):
        # skip expert-marked params (parallel/moe.py): each rank owns its own
        # expert shard, so averaging their grads across ranks would corrupt
        # EP training (reference _ddp_params_and_buffers_to_ignore pattern)
        params = [p for p in self.module.parameters()
                  if p.requires_grad and not getattr(p, "expert", False)]
        cap = int(cap_mb * 1024 * 1024)
        cur, cur_bytes = [], 0
        buckets_params = []
        for p in reversed(params):  # reverse order ~ backward completion order
            cur.append(p)
            cur_bytes += p.numel() * p.element_size()
            if cur_bytes >= cap:
                buckets_params.append(cur)
                cur, cur_bytes = [], 0
        if cur:
            buckets_params.append(cur)
        for group in buckets_params:
            dev = group[0].device
            dt = grad_dtype or group[0].dtype
            b = _Bucket(group, dev, dt)
            self.buckets.append(b)
            for p in group:
                self._param_bucket[id(p)] = b
                h = p.register_post_accumulate_grad_hook(self._hook)
                self._hooks.append(h)

    def _hook(self, p: torch.Tensor):
        if not self._require_sync:
            return
        b = self._param_bucket[id(p)]
        b.ready += 1
        if b.ready == len(b.params):
            self._launch(b)

    def _launch(self, b: _Bucket):
        with torch.no_grad():
            torch._foreach_copy_(b.views, [p.grad for p in b.params])
        b.work = dist.all_reduce(b.flat, op=dist.ReduceOp.SUM, group=self.pg,
                                 async_op=True)

    def finalize(self):
        """Wait for all bucket reductions and write averaged grads back.
        Call after backward(), before optimizer.step()."""
        if not self.enabled or not self._require_sync:
            return
        inv = 1.0 / self.world_size
        for b in self.buckets:
            if b.ready != len(b.params) or b.work is None:
                # straggler bucket (e.g. partially-used model): reduce now
                missing = [p for p in b.params if p.grad is None]
                for p in missing:
                    p.grad = torch.zeros_like(p)
                self._launch(b)
            b.work.wait()
            with torch.no_grad():
                b.flat.mul_(inv)
                torch._foreach_copy_([p.grad for p in b.params], b.views)
            b.ready = 0
            b.work = None

    class _NoSync:
        def __init__(self, outer):
            self.outer = outer

        def __enter__(self):
            self.outer._require_sync = False

        def __exit__(self, *a):
            self.outer._require_sync = True

    def no_sync(self):
        """Skip gradient sync (grad-accumulation inner steps)."""
        return BucketedDataParallel._NoSync(self)

    def forward(self, *args, **kwargs):
        if self.enabled:
            for b in self.buckets:
                b.ready = 0
                b.work = None
        return self.module(*args, **kwargs)

    def state_dict(self, *a, **kw):
        return self.module.state_dict(*a, **kw)

    def load_state_dict(self, *a, **kw):
        return self.module.load_state_dict(*a, **kw)


def wrap_data_parallel(model: nn.Module, style: str = "bucketed", **kw) -> nn.Module:
    """style: 'bucketed' (ours) or 'torch' (torch.nn.parallel.DDP over RCCL)."""
    if not is_dist():
        return model
    if style == "torch":
        from torch.nn.parallel import DistributedDataParallel as DDP

        dev = [torch.cuda.current_device()] if torch.cuda.is_available() else None
        return DDP(model, device_ids=dev, broadcast_buffers=False)
    return BucketedDataParallel(model, **kw)
