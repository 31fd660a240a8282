"""Fused Adam optimizer (K16) with fp32 master weights for bf16 params.

Semantics match torch.optim.Adam (reference main.py:138): m/v moments,
bias correction, optional L2 weight_decay added to the gradient.  On GPU
the per-parameter update is one fused HIP kernel; bf16 params carry an
fp32 master copy updated in fp32 and rounded once per step.

On CPU (torch backend, fp32 params) use torch.optim.Adam directly — this
class is the GPU path.
"""

from __future__ import annotations

from typing import Iterable, Tuple

import torch

from ..ops import functional as Fn
from ..ops import ext as _ext  # noqa: F401  (adam_tick access via Fn.ext)


class FusedAdam:
    def __init__(
        self,
        params: Iterable[torch.nn.Parameter],
        lr: float = 1e-3,
        betas: Tuple[float, float] = (0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.0,
    ) -> None:
        self.params = [p for p in params if p.requires_grad]
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        # device-resident (beta1^t, beta2^t): hipGraph-capturable bias
        # correction (advanced by one adam_tick kernel per step)
        dev = self.params[0].device if self.params else torch.device("cpu")
        self.bc_pow = torch.ones(2, dtype=torch.float64, device=dev)
        self.state = {}
        for p in self.params:
            st = {
                "m": torch.zeros(p.numel(), dtype=torch.float32, device=p.device),
                "v": torch.zeros(p.numel(), dtype=torch.float32, device=p.device),
            }
            if p.dtype == torch.bfloat16:
                st["master"] = p.detach().float().view(-1).clone()
            else:
                st["master"] = None
            self.state[p] = st

    @torch.no_grad()
    def step(self, exclude_ids=None) -> None:
        """One Adam step over params with grads.  ``exclude_ids`` (a set of
        id(param)) skips params whose update will arrive via
        ``step_rows`` in the same logical step (the DDP-overlapped table
        path) — the step counter is bumped here, once."""
        self.step_count += 1
        Fn.ext().adam_tick(self.bc_pow, self.beta1, self.beta2)
        # fp32 params (LN gamma/beta, attention vector, head bias) batch
        # into ONE multi-tensor launch; elementwise Adam makes the result
        # bitwise-identical to per-tensor calls
        f32_batch = ([], [], [], [])
        for p in self.params:
            if p.grad is None or (exclude_ids and id(p) in exclude_ids):
                continue
            st = self.state[p]
            if (st["master"] is None and p.dtype == torch.float32
                    and p.is_cuda and p.grad.dtype == torch.float32
                    and p.grad.is_contiguous() and len(f32_batch[0]) < 8):
                f32_batch[0].append(p.data.view(-1))
                f32_batch[1].append(p.grad.view(-1))
                f32_batch[2].append(st["m"])
                f32_batch[3].append(st["v"])
                continue
            Fn.adam_step(
                p.data, p.grad, st["master"], st["m"], st["v"],
                self.bc_pow, self.lr, self.beta1, self.beta2,
                self.eps, self.weight_decay,
            )
        if f32_batch[0]:
            Fn.ext().adam_step_f32_multi(
                f32_batch[0], f32_batch[1], f32_batch[2], f32_batch[3],
                self.bc_pow, self.lr, self.beta1, self.beta2,
                self.eps, self.weight_decay,
            )

    @torch.no_grad()
    def step_rows(self, p, grad2d, row_lo: int, row_hi: int) -> None:
        """Adam on rows [row_lo, row_hi) of a 2-D param, using a caller-
        owned grad tensor (p.grad may be None).  Uses the step count of
        the enclosing ``step()`` call — elementwise Adam makes the row
        partition exact."""
        W = p.shape[1]
        lo, hi = row_lo * W, row_hi * W
        st = self.state[p]
        Fn.adam_step(
            p.data.view(-1)[lo:hi], grad2d.view(-1)[lo:hi],
            None if st["master"] is None else st["master"][lo:hi],
            st["m"][lo:hi], st["v"][lo:hi],
            self.bc_pow, self.lr, self.beta1, self.beta2,
            self.eps, self.weight_decay,
        )

    def zero_grad(self, set_to_none: bool = False) -> None:
        # NOTE: when grads are BucketedAllReduce views, use ddp.zero_grad()
        # instead — set_to_none=True here would detach the bucket views.
        for p in self.params:
            if p.grad is not None:
                if set_to_none:
                    p.grad = None
                else:
                    p.grad.zero_()
