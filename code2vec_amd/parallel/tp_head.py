"""Vocab-sharded (tensor-parallel) output head + loss — the optional
optimization sketched in SURVEY.md §2.8 ("optional vocab-sharded output
head"): each rank holds an [L/world, E] slice of the label projection, so
the FLOP-dominant output GEMM and its weight gradient shard across GPUs
and the head weight's gradient all-reduce disappears entirely.  Only two
tiny [B]-sized collectives (the online-softmax (max, sum) combines) and
one [B, E] dcv all-reduce cross the wire per step.

This is a self-contained, numerics-verified building block (gloo tests
compare it against the dense reference math bitwise-tolerantly); the
default training path keeps the replicated head — at code2vec's head
sizes (18.5-66 MB) the replicated grad all-reduces overlap backward
(parallel/ddp.py) and DP stays the simpler, hardware-validated route.
Math here is plain torch composed with collectives; the per-shard pieces
map 1:1 onto the existing HIP kernels (head_fwd stats epilogue produces
exactly the local (max, sum) partials this needs).

Reference math being sharded: model/model.py:83 + main.py:251-264.
"""

from __future__ import annotations

import torch
import torch.distributed as dist


def shard_rows(full: torch.Tensor, rank: int, world: int) -> torch.Tensor:
    """Row shard [ceil(L/world)] slice of a [L, ...] tensor (last shard
    may be short)."""
    per = (full.shape[0] + world - 1) // world
    return full[rank * per : (rank + 1) * per]


class VocabParallelHeadLoss(torch.autograd.Function):
    """loss = weighted-NLL(log_softmax(cv @ W^T + b), y) with W/b/weight
    row-sharded over ``group``; returns the global scalar loss.

    Backward produces the LOCAL dW/dbias shards (no collective) and the
    all-reduced dcv.
    """

    @staticmethod
    def forward(ctx, cv, w_shard, b_shard, label, weight_shard, l_offset,
                L, group):
        world = dist.get_world_size(group)
        logits = cv @ w_shard.t().to(cv.dtype) + b_shard.to(cv.dtype)
        lf = logits.float()
        # global online-softmax combine: max then shifted sumexp
        m_loc = lf.max(dim=1).values
        m_glob = m_loc.clone()
        dist.all_reduce(m_glob, op=dist.ReduceOp.MAX, group=group)
        s_loc = torch.exp(lf - m_glob[:, None]).sum(dim=1)
        s_glob = s_loc.clone()
        dist.all_reduce(s_glob, op=dist.ReduceOp.SUM, group=group)
        lse = m_glob + torch.log(s_glob)
        # target logit and target weight live on exactly one rank
        lsh = w_shard.shape[0]
        local = (label >= l_offset) & (label < l_offset + lsh)
        lidx = (label - l_offset).clamp(0, max(lsh - 1, 0))
        logit_y = torch.where(
            local, lf.gather(1, lidx[:, None]).squeeze(1),
            torch.zeros_like(lse))
        w_y = torch.where(
            local, weight_shard[lidx].float(), torch.zeros_like(lse))
        pair = torch.stack([logit_y, w_y])
        dist.all_reduce(pair, op=dist.ReduceOp.SUM, group=group)
        logit_y, w_y = pair[0], pair[1]
        num = (w_y * (lse - logit_y)).sum()
        den = w_y.sum()
        loss = num / den
        ctx.save_for_backward(cv, w_shard, lf, lse, label, w_y, den)
        ctx.meta = (l_offset, lsh, group)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        cv, w_shard, lf, lse, label, w_y, den = ctx.saved_tensors
        l_offset, lsh, group = ctx.meta
        coef = (dloss * w_y / den)[:, None]  # [B, 1]
        g = coef * torch.exp(lf - lse[:, None])  # local softmax block
        local = (label >= l_offset) & (label < l_offset + lsh)
        rows = torch.nonzero(local, as_tuple=True)[0]
        g[rows, (label[rows] - l_offset)] -= coef[rows, 0]
        gt = g.to(w_shard.dtype)
        dw = gt.t() @ cv                      # local shard, NO collective
        dbias = g.sum(dim=0).to(w_shard.dtype)
        dcv = (gt @ w_shard.to(cv.dtype)).float()
        dist.all_reduce(dcv, op=dist.ReduceOp.SUM, group=group)
        return (dcv.to(cv.dtype), dw, dbias, None, None, None, None, None)


def vocab_parallel_head_loss(cv, w_shard, b_shard, label, weight_shard,
                             l_offset, L, group=None):
    return VocabParallelHeadLoss.apply(cv, w_shard, b_shard, label,
                                       weight_shard, l_offset, L, group)
