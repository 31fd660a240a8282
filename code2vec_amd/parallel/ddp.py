"""Gradient all-reduce overlapped with backward (K18) — hybrid scheme.

Three modes per parameter:
- OWNED (the embedding tables, opt-in via ``owned_params``): the table
  grad never goes through autograd accumulation at all.  The embedding
  backward hands the finished grad tensor to a callback (see
  ops/functional.py OWNED_GRAD_KEYS) the moment its cast_clear completes;
  the callback launches CHUNKED async all-reduces on row-ranges of a
  buffer this class owns.  Because autograd never adopts the tensor, the
  round-1 adoption/clone hazard is gone, and the first table's comm
  overlaps the second table's scatter chain.  ``finish_and_step`` then
  waits chunk-by-chunk and runs the fused-Adam row update for each chunk
  while later chunks are still on the wire — the table optimizer work
  (the largest Adam cost) hides most of the exposed all-reduce tail.
- DIRECT (>= direct_threshold bytes — e.g. the output head): the
  parameter's grad tensor is all-reduced in place as soon as its
  post-accumulate hook fires.  p.grad stays None between steps
  (set_to_none), so autograd *assigns* the backward's freshly produced
  grad tensor with zero copies.
- BUCKET (small params: LN, attention, bias, combiner weight): grads live
  as views into a persistent flat buffer; the bucket all-reduces once its
  last grad lands.

Designed for xGMI's per-link ring bound: the big tensors
are each far above RCCL's efficient message size, and the first grads
produced (output head) overlap with the rest of backward.  Comm-cost
model in PERF.md §DP.
"""

from __future__ import annotations

from typing import List, Optional

import torch
import torch.distributed as dist

from ..ops import functional as Fn


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
    """Attach to a model's parameters; between loss.backward() and the
    optimizer step call either ``finish()`` (then ``optim.step()``) or the
    overlapped ``finish_and_step(optim)``."""

    def __init__(
        self,
        params: List[torch.nn.Parameter],
        world_size: int,
        bucket_bytes: int = 32 * 1024 * 1024,
        direct_threshold: int = 1024 * 1024,
        process_group=None,
        enabled: Optional[bool] = None,
        owned_params: Optional[List[torch.nn.Parameter]] = None,
        owned_chunks: int = 4,
    ) -> None:
        self.world_size = world_size
        self.group = process_group
        self.enabled = enabled if enabled is not None else world_size > 1
        self.params = [p for p in params if p.requires_grad]
        self.owned_chunks = owned_chunks

        self.owned_params = [p for p in (owned_params or [])
                             if p.requires_grad]
        owned_ids = set(id(p) for p in self.owned_params)
        rest = [p for p in self.params if id(p) not in owned_ids]

        # owned grads bypass autograd: register the backward callbacks
        self._owned_state = {}   # id(p) -> {"grad", "handles": [(h, view)]}
        self._owned_order = []   # params in grad-completion order
        for p in self.owned_params:
            key = p.data_ptr()
            Fn.OWNED_GRAD_KEYS.add(key)
            Fn.EARLY_GRAD_CALLBACKS[key] = self._make_owned_cb(p)

        self.direct_params = [
            p for p in rest
            if p.numel() * p.element_size() >= direct_threshold
        ]
        bucket_params = [
            p for p in rest
            if p.numel() * p.element_size() < direct_threshold
        ]
        self._direct_set = set(id(p) for p in self.direct_params)
        self._direct_handles = []

        # reverse order ~ autograd completion order; one dtype per bucket
        self.buckets: List[Bucket] = []
        cur: List[torch.nn.Parameter] = []
        cur_bytes = 0
        cur_dtype = None

        def flush():
            nonlocal cur, cur_bytes, cur_dtype
            if cur:
                self.buckets.append(Bucket(cur, cur_dtype, cur[0].device))
            cur, cur_bytes, cur_dtype = [], 0, None

        for p in reversed(bucket_params):
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
                self._param_bucket[id(p)] = b

        self._hooks = []
        if self.enabled:
            for p in rest:
                h = p.register_post_accumulate_grad_hook(self._on_grad_ready)
                self._hooks.append(h)
        self._reset_pending()

    # ------------------------------------------------------------------
    def _make_owned_cb(self, p):
        def cb(grad):
            entry = {"grad": grad, "handles": []}
            if self.enabled:
                rows = grad.shape[0]
                per = (rows + self.owned_chunks - 1) // self.owned_chunks
                for i in range(self.owned_chunks):
                    view = grad[i * per : min((i + 1) * per, rows)]
                    if view.numel() == 0:
                        break
                    h = dist.all_reduce(view, op=dist.ReduceOp.SUM,
                                        group=self.group, async_op=True)
                    entry["handles"].append((h, view))
            self._owned_state[id(p)] = entry
            self._owned_order.append(p)
        return cb

    def _reset_pending(self):
        for b in self.buckets:
            b.pending = len(b.params)
            b.handle = None
        self._direct_handles = []
        self._owned_state = {}
        self._owned_order = []

    def _on_grad_ready(self, param) -> None:
        if id(param) in self._direct_set:
            # NOTE: an "early reduce" of the grad tensor from inside the
            # embedding backward was tried and REVERTED for autograd-
            # accumulated grads: autograd does not reliably adopt the
            # returned tensor (it may clone), so an in-place reduction on
            # the produced tensor can be lost — caught by
            # tests/test_ddp_gpu_gloo.py.  Reduce at hook time; the OWNED
            # path exists precisely to bypass autograd for the tables.
            h = dist.all_reduce(
                param.grad, op=dist.ReduceOp.SUM, group=self.group,
                async_op=True,
            )
            self._direct_handles.append((h, param, param.grad.data_ptr()))
            return
        b = self._param_bucket[id(param)]
        b.pending -= 1
        if b.pending == 0:
            b.handle = dist.all_reduce(
                b.flat, op=dist.ReduceOp.SUM, group=self.group, async_op=True
            )

    # ------------------------------------------------------------------
    def _finish_rest(self) -> None:
        """Wait for the bucket/direct all-reduces and average."""
        if not self.enabled:
            return
        for h, p, gptr in self._direct_handles:
            h.wait()
            if p.grad is None or p.grad.data_ptr() != gptr:
                raise RuntimeError(
                    "direct-reduced gradient tensor changed between the "
                    "hook and finish() — refusing to average the wrong buffer"
                )
            p.grad.div_(self.world_size)
        for b in self.buckets:
            if b.handle is not None:
                b.handle.wait()
                b.flat.div_(self.world_size)
            elif b.pending != len(b.params) and b.pending > 0:
                raise RuntimeError(
                    "BucketedAllReduce.finish(): bucket partially filled — "
                    "backward did not produce all grads in this bucket"
                )
        self._direct_handles = []

    def finish(self) -> None:
        """Non-pipelined completion: wait for everything, publish owned
        grads as ``p.grad`` so a plain ``optimizer.step()`` works."""
        self._finish_rest()
        for p in self.owned_params:
            entry = self._owned_state.get(id(p))
            if entry is None:
                continue
            for h, view in entry["handles"]:
                h.wait()
                view.div_(self.world_size)
            p.grad = entry["grad"]
        self._reset_pending_comm()

    def finish_and_step(self, optim) -> None:
        """Overlapped completion: Adam on everything EXCEPT the owned
        tables runs first (their comm may still be in flight), then each
        owned table updates row-chunk by row-chunk as its all-reduce
        completes — the fused Adam of chunk i executes while chunk i+1 is
        still on the wire."""
        self._finish_rest()
        owned_ids = set(id(p) for p in self.owned_params)
        optim.step(exclude_ids=owned_ids)
        for p in self._owned_order:
            entry = self._owned_state[id(p)]
            grad = entry["grad"]
            if entry["handles"]:
                row0 = 0
                for h, view in entry["handles"]:
                    h.wait()
                    view.div_(self.world_size)
                    optim.step_rows(p, grad, row0, row0 + view.shape[0])
                    row0 += view.shape[0]
            else:
                optim.step_rows(p, grad, 0, grad.shape[0])
        self._reset_pending_comm()

    def _reset_pending_comm(self):
        for b in self.buckets:
            b.pending = len(b.params)
            b.handle = None
        self._direct_handles = []

    def zero_grad(self) -> None:
        for p in self.direct_params:
            p.grad = None  # next backward assigns the fresh tensor
        for p in self.owned_params:
            p.grad = None
        for b in self.buckets:
            b.flat.zero_()
        self._reset_pending()

    def broadcast_parameters(self) -> None:
        if not self.enabled:
            return
        for p in self.params:
            dist.broadcast(p.data, src=0, group=self.group)

    def close(self) -> None:
        """Deregister the owned-grad callbacks (test hygiene: a later
        BucketedAllReduce on the same model replaces them anyway)."""
        for p in self.owned_params:
            key = p.data_ptr()
            Fn.OWNED_GRAD_KEYS.discard(key)
            Fn.EARLY_GRAD_CALLBACKS.pop(key, None)
