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
    def step(self) -> None:
        self.step_count += 1
        for p in self.params:
            if p.grad is None:
                continue
            st = self.state[p]
            Fn.adam_step(
                p.data, p.grad, st["master"], st["m"], st["v"],
                self.step_count, self.lr, self.beta1, self.beta2,
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
