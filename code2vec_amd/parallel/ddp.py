"""Bucketed gradient all-reduce overlapped with backward (K18).

Designed for xGMI's per-link ring bound: gradients live in
persistent flat bucket buffers (p.grad is a view into its bucket, so
autograd accumulates in place with zero copies); as backward finishes the
last grad of a bucket, that bucket's all-reduce launches asynchronously
(RCCL orders it on its own stream) while earlier layers' backward kernels
keep running.  ``finish()`` waits for all handles and averages.

Bucket order follows reverse parameter order (output head first — it is
both the first grad produced and the largest dense matrix, SURVEY.md §7
"Hard parts"), one dtype per bucket.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist


class Bucket:
    def __init__(self, params: List[torch.nn.Parameter], dtype, device):
        self.params = params
        numel = sum(p.numel() for p in params)
        self.flat = torch.zeros(numel, dtype=dtype, device=device)
        self.views = []
        off = 0
        for p in params:
            n = p.numel()
            self.views.append(self.flat[off : off + n].view_as(p))
            off += n
        self.pending = 0
        self.handle = None


class BucketedAllReduce:
    """Attach to a model's parameters; call ``finish()`` between
    loss.backward() and optimizer.step()."""

    def __init__(
        self,
        params: List[torch.nn.Parameter],
        world_size: int,
        bucket_bytes: int = 32 * 1024 * 1024,
        process_group=None,
        enabled: Optional[bool] = None,
    ) -> None:
        self.world_size = world_size
        self.group = process_group
        self.enabled = enabled if enabled is not None else world_size > 1
        self.params = [p for p in params if p.requires_grad]

        # reverse order ~ autograd completion order
        self.buckets: List[Bucket] = []
        cur: List[torch.nn.Parameter] = []
        cur_bytes = 0
        cur_dtype = None

        def flush():
            nonlocal cur, cur_bytes, cur_dtype
            if cur:
                self.buckets.append(Bucket(cur, cur_dtype, cur[0].device))
            cur, cur_bytes, cur_dtype = [], 0, None

        for p in reversed(self.params):
            nbytes = p.numel() * p.element_size()
            if cur and (p.dtype != cur_dtype or cur_bytes + nbytes > bucket_bytes):
                flush()
            cur.append(p)
            cur_dtype = p.dtype
            cur_bytes += nbytes
        flush()

        self._param_bucket = {}
        for b in self.buckets:
            for p, view in zip(b.params, b.views):
                p.grad = view  # autograd accumulates directly into the bucket
                self._param_bucket[p] = b

        self._hooks = []
        if self.enabled:
            for p in self.params:
                h = p.register_post_accumulate_grad_hook(self._on_grad_ready)
                self._hooks.append(h)
        self._reset_pending()

    def _reset_pending(self):
        for b in self.buckets:
            b.pending = len(b.params)
            b.handle = None

    def _on_grad_ready(self, param) -> None:
        b = self._param_bucket[param]
        b.pending -= 1
        if b.pending == 0:
            b.handle = dist.all_reduce(
                b.flat, op=dist.ReduceOp.SUM, group=self.group, async_op=True
            )

    def finish(self) -> None:
        """Wait for outstanding all-reduces and average; call every step."""
        if not self.enabled:
            return
        for b in self.buckets:
            if b.handle is not None:
                b.handle.wait()
                b.flat.div_(self.world_size)
            elif b.pending != len(b.params) and b.pending > 0:
                raise RuntimeError(
                    "BucketedAllReduce.finish(): bucket partially filled — "
                    "backward did not produce all grads in this bucket"
                )
        self._reset_pending()

    def zero_grad(self) -> None:
        for b in self.buckets:
            b.flat.zero_()
        self._reset_pending()

    def broadcast_parameters(self) -> None:
        if not self.enabled:
            return
        for p in self.params:
            dist.broadcast(p.data, src=0, group=self.group)
