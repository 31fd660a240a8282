"""Autograd wrappers around the HIP kernels.

Each ``torch.autograd.Function`` here pairs a hand-written CDNA4 forward
kernel with its backward kernels.  Nearly every
hot GEMM ended up custom after A/B against the tuned libraries: the
combiner forward (fused LN/tanh/dropout epilogue), combiner dgrad
(dgrad2) and wgrad (split-K — the skinny big-K shape where hipBLASLt is
~3.5x off), the head forward (+online-softmax stats epilogue) and the
fused recompute-G head backward (dW/dbias/dcv).  rocBLAS/hipBLASLt
remain only where they measured faster or equal: the head forward above
the vocab gate with C2V_HF_AIMG=0, and small reductions torch already
does at bandwidth (the wgrad partial-slab sum).

Numerical contracts are defined by ops/reference.py; tests compare against
it in fp32.
"""

from __future__ import annotations

import os
from typing import Optional

import torch

# epilogue store scheme for combiner_fwd: 0 = per-element stores,
# 1 = per-wave LDS bounce + coalesced 16-B flush (kbench-selected)
_FWD_EPI = int(os.environ.get("C2V_FWD_EPI", "1"))
_SCATTER_R = int(os.environ.get("C2V_SCATTER_R", "16"))
# Fused gather+combiner forward, opt-in via C2V_FUSE=1.  Measured OFF-better
# on top11 (1.415 vs 1.478 ms/step): fusing saves the 61 us gather kernel but
# costs +16 us in combiner_fwd (scattered row loads inside the MFMA loop) and
# +109 us in wgrad (GATHER=1 re-gathers X every split-K pass), net +64 us.
# The split gather kernel streams at HBM BW, so there is no bandwidth win to
# recover.  Kept (correct, GPU-tested) for small-table configs where the
# gather output (B*C*KP bf16) no longer fits comfortably.
FUSE_GATHER_COMBINER = os.environ.get("C2V_FUSE") == "1"
# custom output-head forward (+ fused loss statistics); C2V_HEAD_FWD=0
# falls back to hipBLASLt linear + full-pass loss forward.  Measured
# faster through the derived top11 vocab (with normal stores + the fused
# backward: 1.403 vs 1.412 ms/step at L=72,416) but still slower at
# java-large's L=261k (strided W fragment reads stream from HBM), so
# vocabs past C2V_HEAD_FWD_MAXL take the library path.
_HEAD_FWD = os.environ.get("C2V_HEAD_FWD", "1") == "1"
_HEAD_FWD_MAXL = int(os.environ.get("C2V_HEAD_FWD_MAXL", "98304"))
# above MAXL: custom head forward consuming a swizzle_a W fragment image
# (contiguous A reads; re-enables the fused stats epilogue at java-large
# scale).  C2V_HF_AIMG=0 falls back to hipBLASLt + lsm_partial there.
_HF_AIMG = os.environ.get("C2V_HF_AIMG", "1") == "1"
# hipCUB decoupled-lookback scan for the counting-sort cursor (one launch
# vs the 3-kernel partials/spine/apply chain); C2V_SCAN_CUB=0 reverts
_SCAN_CUB = os.environ.get("C2V_SCAN_CUB", "1") == "1"
_CUB_TEMP_BYTES: dict = {}
# streaming head backward: recompute logits by MFMA inside dcv instead of
# reading the [B, L] tensor (C2V_HB_RC=0 reverts to the logits-reading
# kernel)
_HB_RC = os.environ.get("C2V_HB_RC", "0") == "1"
# custom split-K dcv in the head backward (C2V_HEAD_DGRAD=0 -> rocBLAS)
_HEAD_DGRAD = os.environ.get("C2V_HEAD_DGRAD", "1") == "1"
# combiner dgrad through dgrad2.hip (C2V_DGRAD2=0 -> rocBLAS)
_DGRAD2 = os.environ.get("C2V_DGRAD2", "1") == "1"
# fused head+loss backward (recompute-G; dlogits never materialized).
# C2V_FUSED_HEAD=0 falls back to the unfused OutputHead+FusedLogSoftmaxNLL
# chain (kept for A/B and as the general-shape path).
FUSED_HEAD = os.environ.get("C2V_FUSED_HEAD", "1") == "1"
_NONE_T = torch.Tensor()  # "not provided" sentinel for optional kernel args

# Owner-buffer gradient flow for the embedding tables (parallel/ddp.py):
# when a param's data_ptr is in OWNED_GRAD_KEYS, the embedding backward
# hands its finished grad tensor to EARLY_GRAD_CALLBACKS[key] the moment
# its cast_clear completes and returns None to autograd.  Autograd then
# never adopts (and possibly CLONES) the tensor — the round-1 hazard that
# made in-place early all-reduce unsafe — so the DDP layer can launch
# chunked async all-reduces on a buffer it owns while the REST of backward
# (the other table's sort/scatter chain) still runs.
EARLY_GRAD_CALLBACKS = {}
OWNED_GRAD_KEYS = set()

from . import ext, round_up

_rng_state = {"seed": None, "dev_off": {}}


def _philox_state(device):
    """(seed, device offset scalar) for the counter-based dropout RNG.
    The offset lives in DEVICE memory and is advanced by the consuming
    kernel launch chain, so a hipGraph-captured training step keeps
    drawing fresh masks on every replay."""
    if _rng_state["seed"] is None:
        _rng_state["seed"] = torch.initial_seed() & 0x7FFFFFFFFFFFFFFF
    key = str(device)
    off = _rng_state["dev_off"].get(key)
    if off is None:
        off = torch.zeros(1, dtype=torch.int64, device=device)
        _rng_state["dev_off"][key] = off
    return _rng_state["seed"], off


def reseed_dropout_rng(seed: int) -> None:
    _rng_state["seed"] = seed & 0x7FFFFFFFFFFFFFFF
    for off in _rng_state["dev_off"].values():
        off.zero_()


class GatherConcat(torch.autograd.Function):
    """K1/K2 + K13: fused triple embedding gather+concat; scatter-add bwd.

    out[b*C+c] = [term[starts] (TS cols) | path[paths] (PS) | term[ends] (TS)]
    """

    @staticmethod
    def forward(ctx, starts, paths, ends, term_w, path_w):
        B, C = starts.shape
        TS = term_w.shape[1]
        PS = path_w.shape[1]
        out = torch.empty(
            B * C, round_up(2 * TS + PS, 32), dtype=torch.bfloat16,
            device=starts.device
        )
        ext().gather_concat_fwd(starts, paths, ends, term_w, path_w, out)
        ctx.save_for_backward(starts, paths, ends)
        ctx.shapes = (term_w.shape, path_w.shape)
        ctx.param_keys = (term_w.data_ptr(), path_w.data_ptr())
        return out

    @staticmethod
    def backward(ctx, grad_out):
        starts, paths, ends = ctx.saved_tensors
        term_shape, path_shape = ctx.shapes
        dterm, dpath = _scatter_embedding_grads(
            starts, paths, ends, grad_out.contiguous(), term_shape,
            path_shape, ctx.param_keys[0], ctx.param_keys[1]
        )
        return None, None, None, dterm, dpath


def _scatter_embedding_grads(starts, paths, ends, gout, term_shape,
                             path_shape, term_key=None, path_key=None):
    """K13 v4: sort-based segmented scatter of a [M, KP] grad into bf16
    dense embedding grads.  Counting-sort groups the index lists; run-owner
    waves write interior rows' bf16 grads directly; boundary-crossing runs
    (heavy hitters) combine via flagged persistent fp32 scratch in
    cast_clear_rows (which also re-zeroes what it consumed)."""
    dev = gout.device
    M = starts.numel()
    TS, PS = term_shape[1], path_shape[1]
    KP = gout.shape[1]
    dterm32 = _scratch_f32("term", term_shape, dev)
    dpath32 = _scratch_f32("path", path_shape, dev)
    flags_t = _scratch_flags("term", term_shape[0], dev)
    flags_p = _scratch_flags("path", path_shape[0], dev)
    dterm = torch.empty(term_shape, dtype=torch.bfloat16, device=dev)
    dpath = torch.empty(path_shape, dtype=torch.bfloat16, device=dev)
    # term's FULL chain (sort+scatter+cast) runs before path's STARTS, so
    # the early callback can put term's all-reduce on the wire while
    # path's ~100 us of scatter kernels still execute (DP overlap)
    idx_se = torch.cat([starts.view(-1), ends.view(-1)])
    sorted_se, perm_se, counts_se = _group_by_index(idx_se, term_shape[0],
                                                    pool_tag="term")
    ext().embed_scatter_sorted(sorted_se, perm_se, gout, dterm32, dterm,
                               flags_t, M, KP, 0, TS + PS, _SCATTER_R)
    ext().cast_clear_rows(dterm32, counts_se, flags_t, dterm)
    if term_key is not None:
        cb = EARLY_GRAD_CALLBACKS.get(term_key)
        if cb is not None:
            cb(dterm)
    sorted_p, perm_p, counts_p = _group_by_index(paths.view(-1),
                                                 path_shape[0],
                                                 pool_tag="path")
    ext().embed_scatter_sorted(sorted_p, perm_p, gout, dpath32, dpath,
                               flags_p, M, KP, TS, TS, _SCATTER_R)
    ext().cast_clear_rows(dpath32, counts_p, flags_p, dpath)
    if path_key is not None:
        cb = EARLY_GRAD_CALLBACKS.get(path_key)
        if cb is not None:
            cb(dpath)
    if term_key in OWNED_GRAD_KEYS:
        dterm = None  # the callback owner keeps the buffer; autograd must
    if path_key in OWNED_GRAD_KEYS:  # not adopt (or clone) it
        dpath = None
    return dterm, dpath


_scratch_cache = {}


def _scratch_f32(tag: str, shape, device) -> torch.Tensor:
    """Persistent fp32 scatter scratch (invariant: all-zero between steps —
    maintained by cast_clear_rows).  ``tag`` keeps the term/path buffers
    distinct even when both tables have identical shapes."""
    key = (tag, tuple(shape), str(device))
    buf = _scratch_cache.get(key)
    if buf is None:
        buf = torch.zeros(shape, dtype=torch.float32, device=device)
        _scratch_cache[key] = buf
    return buf


def _scratch_bf16(tag: str, shape, device) -> torch.Tensor:
    """Persistent bf16 scratch (fully overwritten by its producer)."""
    key = ("bf16", tag, tuple(shape), str(device))
    buf = _scratch_cache.get(key)
    if buf is None:
        buf = torch.empty(shape, dtype=torch.bfloat16, device=device)
        _scratch_cache[key] = buf
    return buf


def _scratch_i32(tag: str, n: int, device, zero: bool = True) -> torch.Tensor:
    """Persistent int32 scratch; ``zero=True`` buffers carry an all-zero
    invariant between steps (maintained by their consumer)."""
    key = ("i32", tag, n, str(device))
    buf = _scratch_cache.get(key)
    if buf is None:
        maker = torch.zeros if zero else torch.empty
        buf = maker(n, dtype=torch.int32, device=device)
        _scratch_cache[key] = buf
    return buf


def _scratch_u8(tag: str, n: int, device) -> torch.Tensor:
    key = ("u8", tag, n, str(device))
    buf = _scratch_cache.get(key)
    if buf is None:
        buf = torch.empty(n, dtype=torch.uint8, device=device)
        _scratch_cache[key] = buf
    return buf


def _scratch_flags(tag: str, rows: int, device) -> torch.Tensor:
    key = ("flags", tag, rows, str(device))
    buf = _scratch_cache.get(key)
    if buf is None:
        buf = torch.zeros(rows, dtype=torch.uint8, device=device)
        _scratch_cache[key] = buf
    return buf


def _combiner_dgrad(dz: torch.Tensor, w: torch.Tensor) -> torch.Tensor:
    """dx = dz @ w ([M, EP] @ [EP, KP], K = EP = 128) — a 131-MB
    streaming write that hipBLASLt runs 3x off its floor (MT64x256x64,
    91 us at top11).  dgrad2.hip computes it transposed-orientation with
    the weight re-transposed to [KP, EP]: both fragments contiguous,
    dz tiles staged block-cooperatively, packed full-line nontemporal
    stores; 128-row batch tiles keep per-wave overhead half of a naive
    head_fwd reuse (which measured a wash — see PERF.md)."""
    M, EP = dz.shape
    KP = w.shape[1]
    if not (_DGRAD2 and dz.is_cuda and EP == 128 and KP % 8 == 0):
        return dz @ w
    w2 = w.t().contiguous()  # [KP, EP], 82 KB at top11
    dx = torch.empty(M, KP, dtype=torch.bfloat16, device=w.device)
    ext().dgrad2(dz, w2, dx)
    return dx


def _slab_sum(p: torch.Tensor) -> torch.Tensor:
    """Sum reduction-partial slabs [S, N] -> [N] f32 (block-per-column
    kernel; torch's generic reduce costs ~10 us per call at these tiny
    shapes)."""
    out = torch.empty(p.shape[-1], dtype=torch.float32, device=p.device)
    ext().slab_sum_f32(p, out)
    return out


def _group_by_index(idx: torch.Tensor, table_rows: int,
                    pool_tag: Optional[str] = None):
    """Counting sort: returns (sorted_idx i32, perm i64, counts i32)
    grouping equal indexes contiguously (ascending).

    With ``pool_tag`` (the training hot path) the histogram lives in a
    PERSISTENT all-zero buffer whose invariant cast_clear_rows restores
    by consuming the counts it reads, and the cursor comes from the
    custom 3-kernel exclusive scan — this removes two torch.zeros fills
    and a rocprim lookback scan (+init) per table per step (~21 us each).
    Callers that do NOT route the counts through cast_clear_rows must use
    the default (fresh-buffer) form.
    """
    N = idx.numel()
    dev = idx.device
    if pool_tag is not None:
        counts = _scratch_i32(f"counts_{pool_tag}", table_rows + 1, dev)
        cursor = _scratch_i32(f"cursor_{pool_tag}", table_rows + 1, dev,
                              zero=False)
        spart = _scratch_i32(f"scanp_{pool_tag}",
                             (table_rows + 1 + 1023) // 1024, dev,
                             zero=False)
    else:
        counts = torch.zeros(table_rows + 1, dtype=torch.int32, device=dev)
        cursor = None
    empty_i = torch.empty(0, dtype=torch.int32, device=dev)
    empty_l = torch.empty(0, dtype=torch.int64, device=dev)
    ext().group_by_index(idx, counts, empty_i, empty_i, empty_l, True)
    if pool_tag is not None:
        if _SCAN_CUB:
            tb = _CUB_TEMP_BYTES.get(table_rows + 1)
            if tb is None:
                tb = int(ext().cub_scan_temp_bytes(table_rows + 1))
                _CUB_TEMP_BYTES[table_rows + 1] = tb
            temp = _scratch_u8(f"cubscan_{pool_tag}", max(tb, 16), dev)
            ext().cub_exclusive_scan(counts, temp, cursor)
        else:
            ext().exclusive_scan(counts, spart, cursor)
    else:
        cursor = torch.zeros_like(counts)
        cursor[1:] = torch.cumsum(counts[:-1], 0)
    sorted_idx = torch.empty(N, dtype=torch.int32, device=dev)
    perm = torch.empty(N, dtype=torch.int64, device=dev)
    ext().group_by_index(idx, counts, cursor, sorted_idx, perm, False)
    return sorted_idx, perm, counts


class CombinerLNTanh(torch.autograd.Function):
    """K3-K6: MFMA GEMM (x @ w) -> LayerNorm(E) -> tanh -> dropout, fused.

    x: bf16 [M, KP]; w: bf16 [EP, KP] (TRANSPOSED so the MFMA B fragment
    is memory-contiguous); gamma/beta: f32 [EP]; E = valid cols.
    Saves z (pre-LN GEMM output, bf16) + per-row mean/rstd for backward;
    the dropout mask is recomputed from the counter RNG, never stored.
    """

    @staticmethod
    def forward(ctx, x, w, gamma, beta, E: int, p: float, training: bool):
        M = x.shape[0]
        EP = w.shape[0]
        out = torch.empty(M, EP, dtype=torch.bfloat16, device=x.device)
        z = torch.empty(M, EP, dtype=torch.bfloat16, device=x.device)
        mean = torch.empty(M, dtype=torch.float32, device=x.device)
        rstd = torch.empty(M, dtype=torch.float32, device=x.device)
        p_eff = float(p) if training else 0.0
        seed, off_t = (_philox_state(x.device) if p_eff > 0.0
                       else (0, _philox_state(x.device)[1]))
        ext().combiner_fwd(x, w, gamma, beta, out, z, mean, rstd, E, p_eff,
                           seed, off_t, _FWD_EPI)
        ctx.save_for_backward(x, w, gamma, beta, z, mean, rstd, out)
        ctx.meta = (E, p_eff)
        return out

    @staticmethod
    def backward(ctx, dout):
        x, w, gamma, beta, z, mean, rstd, out = ctx.saved_tensors
        E, p = ctx.meta
        M, EP = z.shape
        dz = torch.empty(M, EP, dtype=torch.bfloat16, device=z.device)
        # per-block partials (kernel grid cap = 1024): summed here, which is
        # deterministic and avoids same-address atomic serialization
        nblocks = min((M + 4 - 1) // 4, 1024)
        dgamma_p = torch.empty(nblocks, EP, dtype=torch.float32, device=z.device)
        dbeta_p = torch.empty(nblocks, EP, dtype=torch.float32, device=z.device)
        ext().combiner_bwd(
            dout.contiguous(), z, out, mean, rstd, gamma, beta, dz, dgamma_p,
            dbeta_p, E, p,
        )
        dgamma = _slab_sum(dgamma_p)
        dbeta = _slab_sum(dbeta_p)
        # dgrad: plain GEMM -> rocBLAS (TunableOp-tuned).  A hand-written
        # direct-fragment kernel (ops/csrc/dgrad.hip) measured 192 us vs
        # rocBLAS's 64 us at the top11 shape (8-wave full-KP duplicates B
        # L2 traffic) and is kept only as a reference; re-enable via
        # C2V_CUSTOM_DGRAD=1 for experiments.
        if os.environ.get("C2V_CUSTOM_DGRAD") == "1":
            w2 = w.t().contiguous()
            dx = torch.empty(M, w.shape[1], dtype=torch.bfloat16,
                             device=w.device)
            ext().dgrad(dz, w2, dx)
        else:
            dx = _combiner_dgrad(dz, w)
        # wgrad: custom split-K MFMA kernel for the skinny big-K shape
        # (hipBLASLt is ~3.5x off there); partial slabs summed here.
        KP = x.shape[1]
        if EP <= 128 and KP <= 512:
            nsplit = 256
            partials = torch.empty(nsplit, KP, EP, dtype=torch.float32,
                                   device=x.device)
            ext().wgrad(x, dz, partials)
            dw = partials.sum(dim=0).t().contiguous().to(torch.bfloat16)
        else:
            dw = (x.t() @ dz).t().contiguous()
        return dx, dw, dgamma, dbeta, None, None, None


class FusedGatherCombiner(torch.autograd.Function):
    """K1-K6 fully fused: embedding gather + MFMA combiner GEMM + LN + tanh
    + dropout in ONE kernel — the [M, KP] concat tensor is never
    materialized on the forward path.  Backward: fused LN/tanh/dropout
    chain -> dz; dW via the re-gathering split-K wgrad; dX (for the
    embedding scatter) via one rocBLAS GEMM; embedding grads via the
    sort-based scatter."""

    @staticmethod
    def forward(ctx, starts, paths, ends, term_w, path_w, w, gamma, beta,
                E: int, p: float, training: bool):
        M = starts.numel()
        EP, KP = w.shape
        dev = starts.device
        out = torch.empty(M, EP, dtype=torch.bfloat16, device=dev)
        z = torch.empty(M, EP, dtype=torch.bfloat16, device=dev)
        mean = torch.empty(M, dtype=torch.float32, device=dev)
        rstd = torch.empty(M, dtype=torch.float32, device=dev)
        p_eff = float(p) if training else 0.0
        seed, off_t = (_philox_state(dev) if p_eff > 0.0
                       else (0, _philox_state(dev)[1]))
        ext().gather_combiner_fwd(starts, paths, ends, term_w, path_w, w,
                                  gamma, beta, out, z, mean, rstd, KP, E,
                                  p_eff, seed, off_t)
        ctx.save_for_backward(starts, paths, ends, term_w, path_w, w, gamma,
                              beta, z, mean, rstd, out)
        ctx.meta = (E, p_eff)
        return out

    @staticmethod
    def backward(ctx, dout):
        (starts, paths, ends, term_w, path_w, w, gamma, beta, z, mean, rstd,
         out) = ctx.saved_tensors
        E, p = ctx.meta
        M, EP = z.shape
        KP = w.shape[1]
        dz = torch.empty(M, EP, dtype=torch.bfloat16, device=z.device)
        nblocks = min((M + 4 - 1) // 4, 1024)
        dgamma_p = torch.empty(nblocks, EP, dtype=torch.float32, device=z.device)
        dbeta_p = torch.empty(nblocks, EP, dtype=torch.float32, device=z.device)
        ext().combiner_bwd(dout.contiguous(), z, out, mean, rstd, gamma,
                           beta, dz, dgamma_p, dbeta_p, E, p)
        dgamma = _slab_sum(dgamma_p)
        dbeta = _slab_sum(dbeta_p)
        # dW: re-gathering split-K wgrad (X is never materialized)
        if EP <= 128 and KP <= 512:
            partials = torch.empty(256, KP, EP, dtype=torch.float32,
                                   device=z.device)
            ext().wgrad_gather(starts, paths, ends, term_w, path_w, dz,
                               partials, KP)
            dw = partials.sum(dim=0).t().contiguous().to(torch.bfloat16)
        else:
            x = GatherConcat.apply(starts, paths, ends, term_w, path_w)
            dw = (x.t() @ dz).t().contiguous()
        # dX for the embedding scatter
        dx = _combiner_dgrad(dz, w)
        dterm, dpath = _scatter_embedding_grads(
            starts, paths, ends, dx, term_w.shape, path_w.shape,
            term_w.data_ptr(), path_w.data_ptr()
        )
        return (None, None, None, dterm, dpath, dw, dgamma, dbeta,
                None, None, None)


class AttentionPool(torch.autograd.Function):
    """K7-K9 fused: masked single-query attention + weighted pool.

    ccv: bf16 [B, C, EP]; a: f32 [EP]; starts: i32 [B, C] (mask = starts>0).
    Returns cv f32 [B, EP] and attn f32 [B, C].
    """

    @staticmethod
    def forward(ctx, ccv, a, starts, E: int):
        B, C, EP = ccv.shape
        cv = torch.empty(B, EP, dtype=torch.float32, device=ccv.device)
        attn = torch.empty(B, C, dtype=torch.float32, device=ccv.device)
        ext().attention_fwd(ccv, a, starts, cv, attn, E)
        ctx.save_for_backward(ccv, a, starts, attn)
        ctx.E = E
        ctx.set_materialize_grads(False)
        return cv, attn

    @staticmethod
    def backward(ctx, dcv, dattn):
        ccv, a, starts, attn = ctx.saved_tensors
        B, C, EP = ccv.shape
        if dcv is None:
            dcv = torch.zeros(B, EP, dtype=torch.float32, device=ccv.device)
        dccv = torch.empty_like(ccv)
        # per-block partials [B, EP], summed here (deterministic, no atomics)
        da_p = torch.empty(B, EP, dtype=torch.float32, device=ccv.device)
        has_dattn = dattn is not None
        if not has_dattn:
            dattn = torch.empty(0, dtype=torch.float32, device=ccv.device)
        ext().attention_bwd(
            dcv.contiguous(), dattn.contiguous() if has_dattn else dattn,
            ccv, a, starts, attn, dccv, da_p, ctx.E, has_dattn,
        )
        return dccv, _slab_sum(da_p), None, None


class OutputHead(torch.autograd.Function):
    """K10: label projection logits = cv @ W^T + b with a custom backward:
    dW/dcv via rocBLAS, dbias via the colsum kernel (torch's reduce is ~4x
    off HBM bandwidth on [B, 30k+] bf16)."""

    @staticmethod
    def forward(ctx, cv_bf16, w, bias):
        B, EP = cv_bf16.shape
        L = w.shape[0]
        if (_HEAD_FWD and cv_bf16.is_cuda and EP % 32 == 0
                and L <= _HEAD_FWD_MAXL):
            # custom MFMA forward: both fragments are contiguous in memory
            # (K = EP is the fast axis of both cv and w), so this streams at
            # the C-write bound where hipBLASLt runs a K=128 GEMM pipeline.
            # When grads are on (training), the epilogue also emits the
            # per-row online-softmax partials the NLL loss consumes, saving
            # the loss kernel's full re-read of logits.
            logits = torch.empty(B, L, dtype=torch.bfloat16,
                                 device=cv_bf16.device)
            if cv_bf16.requires_grad or w.requires_grad:
                gx = (L + 255) // 256  # one partial per 256-label block
                pm = torch.empty(gx, B, dtype=torch.float32, device=w.device)
                ps = torch.empty_like(pm)
            else:
                pm = ps = _NONE_T
            ext().head_fwd(cv_bf16, w, bias.float(), logits, pm, ps)
            if pm is not _NONE_T:
                logits._c2v_lsm_partials = (pm, ps)
        else:
            logits = torch.nn.functional.linear(
                cv_bf16, w, bias.to(torch.bfloat16))
        ctx.save_for_backward(cv_bf16, w)
        return logits

    @staticmethod
    def backward(ctx, dlogits):
        cv, w = ctx.saved_tensors
        dlogits = dlogits.contiguous()
        B, L = dlogits.shape
        EP = cv.shape[1]
        if (_HEAD_DGRAD and EP == 128 and L % 8 == 0 and dlogits.is_cuda
                and L <= _HEAD_FWD_MAXL):  # big L: slab traffic dominates
            # split-K MFMA dcv (head_dgrad.hip): hipBLASLt runs this
            # skinny-output huge-K GEMM ~6x off the traffic floor.  W is
            # transposed once so both fragments are contiguous loads.
            wt = _scratch_bf16("head_wt", (128, L), w.device)
            ext().transpose_w(w, wt)
            split = (L + 511) // 512
            partials = _scratch_f32("head_dgrad", (split, B, 128), w.device)
            ext().head_dgrad(dlogits, wt, partials)
            dcv = torch.empty(B, 128, dtype=torch.bfloat16,
                              device=w.device)
            ext().slab_sum_bf16(partials, dcv)
        else:
            dcv = dlogits @ w                  # [B, EP] bf16 (rocBLAS)
        # default OFF: measured slower than TunableOp rocBLAS at the
        # top11 shape (single-buffered staging; see PERF.md)
        if (os.environ.get("C2V_HEAD_WGRAD", "0") == "1"
                and B % 32 == 0 and L % 8 == 0 and EP % 32 == 0
                and EP <= 128):
            dw = torch.empty(L, EP, dtype=torch.bfloat16, device=w.device)
            ext().head_wgrad(dlogits, cv, dw)
        else:
            dw = dlogits.t() @ cv              # [L, EP] bf16 (rocBLAS)
        dbias = torch.zeros(w.shape[0], dtype=torch.float32,
                            device=w.device)
        ext().colsum_bf16(dlogits, dbias)
        return dcv, dw, dbias


def _acc_reduce(acc_part):
    """Deterministic (fixed-order) reduction of per-block loss partials
    [S, 2] -> acc[2] = (sum w_y*nll, sum w_y).  The previous atomic-add
    accumulation was ordering-nondeterministic at the fp32 ulp level and
    made identical runs diverge after two steps of bf16 rounding."""
    acc = torch.empty(2, dtype=torch.float32, device=acc_part.device)
    ext().slab_sum_f32(acc_part, acc)
    return acc


def _finalize_stats(logits, pm, ps, label, weight, lse):
    B = logits.shape[0]
    acc_p = torch.empty((B + 15) // 16, 2, dtype=torch.float32,
                        device=logits.device)
    ext().logsoftmax_nll_finalize(logits, pm, ps, label, weight, lse, acc_p)
    return _acc_reduce(acc_p)


def _lsm_stats(logits, label, weight, lse):
    """Fill lse and return acc from a full pass over logits.  Large L
    takes the 2-D-grid partials kernel + finalize (the one-block-per-row
    walk is latency-bound at ~3.6 TB/s at L = 261k: 150 -> ~75 us)."""
    B, L = logits.shape
    if L >= 32768:
        gx = (L + 16383) // 16384
        pm = torch.empty(gx, B, dtype=torch.float32, device=logits.device)
        ps = torch.empty_like(pm)
        ext().lsm_partial(logits, pm, ps)
        return _finalize_stats(logits, pm, ps, label, weight, lse)
    acc_p = torch.empty(B, 2, dtype=torch.float32, device=logits.device)
    ext().logsoftmax_nll_fwd(logits, label, weight, lse, acc_p)
    return _acc_reduce(acc_p)


class FusedLogSoftmaxNLL(torch.autograd.Function):
    """K12: full-vocab log-softmax + weighted NLL, fused fwd and bwd.

    logits: bf16 [B, L]; label: i64 [B]; weight: f32 [L] (1/freq weights,
    reference main.py:129-130).  loss = sum(w_y*(lse - logit_y)) / sum(w_y).
    """

    @staticmethod
    def forward(ctx, logits, label, weight):
        B, L = logits.shape
        lse = torch.empty(B, dtype=torch.float32, device=logits.device)
        # acc[0] = sum(w_y * nll), acc[1] = sum(w_y)
        partials = getattr(logits, "_c2v_lsm_partials", None)
        if partials is not None:
            # the head-forward epilogue already reduced per-row (max, sum)
            # partials over the logits — merge them instead of re-reading
            # the whole [B, L] matrix
            pm, ps = partials
            del logits._c2v_lsm_partials
            acc = _finalize_stats(logits, pm, ps, label, weight, lse)
        else:
            acc = _lsm_stats(logits, label, weight, lse)
        loss = acc[0] / acc[1]
        ctx.save_for_backward(logits, label, weight, lse, acc)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, label, weight, lse, acc = ctx.saved_tensors
        dlogits = torch.empty_like(logits)
        ext().logsoftmax_nll_bwd(
            logits, label, weight, lse, acc, dloss.contiguous().float(), dlogits
        )
        return dlogits, None, None


class AngularMarginHead(torch.autograd.Function):
    """K11: ArcFace-style head fully on HIP kernels (angular.hip).

    Forward: per-row inverse norms -> unit matrices -> MFMA cosine GEMM
    with the margin epilogue (emits scaled outputs AND the raw cosine for
    backward).  Backward: dcos elementwise kernel; du = dcos @ U_w via the
    split-K head_dgrad kernel; dv = dcos^T @ U_cv via head_wgrad; both
    projected through the normalize backward (norm_project kernel).
    Reference math: model/model.py:71-80 (th/mm computed-but-unused there).
    EP = 128 only; callers gate and fall back to ops/reference.py.
    """

    @staticmethod
    def forward(ctx, cv_bf16, w, label, cos_m, sin_m, s):
        B = cv_bf16.shape[0]
        L = w.shape[0]
        dev = w.device
        inv_c = torch.empty(B, dtype=torch.float32, device=dev)
        inv_w = torch.empty(L, dtype=torch.float32, device=dev)
        ext().inv_rownorm(cv_bf16, inv_c)
        ext().inv_rownorm(w, inv_w)
        ucv = torch.empty_like(cv_bf16)
        uw = _scratch_bf16("ang_uw", (L, 128), dev)
        ext().rowscale(cv_bf16, inv_c, ucv)
        ext().rowscale(w, inv_w, uw)
        out = torch.empty(B, L, dtype=torch.bfloat16, device=dev)
        cos = torch.empty(B, L, dtype=torch.bfloat16, device=dev)
        ext().angular_fwd(ucv, uw, label, out, cos, cos_m, sin_m, s)
        ctx.save_for_backward(ucv, uw, inv_c, inv_w, cos, label)
        ctx.consts = (cos_m, sin_m, s)
        return out

    @staticmethod
    def backward(ctx, dout):
        ucv, uw, inv_c, inv_w, cos, label = ctx.saved_tensors
        cos_m, sin_m, s = ctx.consts
        B, L = cos.shape
        dev = cos.device
        dcos = torch.empty_like(cos)
        ext().angular_dcos(dout.contiguous().to(torch.bfloat16), cos, label,
                           dcos, cos_m, sin_m, s)
        # du = dcos @ U_w  (split-K head_dgrad; rocBLAS fallback off-shape)
        if L % 8 == 0:
            uwt = _scratch_bf16("ang_uwt", (128, L), dev)
            ext().transpose_w(uw, uwt)
            split = (L + 511) // 512
            partials = _scratch_f32("ang_dcv", (split, B, 128), dev)
            ext().head_dgrad(dcos, uwt, partials)
            du = torch.empty(B, 128, dtype=torch.bfloat16, device=dev)
            ext().slab_sum_bf16(partials, du)
        else:
            du = dcos @ uw
        # dv = dcos^T @ U_cv  (head_wgrad; rocBLAS fallback off-shape)
        dv = torch.empty(L, 128, dtype=torch.bfloat16, device=dev)
        if B % 32 == 0 and L % 8 == 0:
            ext().head_wgrad(dcos, ucv, dv)
        else:
            dv.copy_(dcos.t() @ ucv)
        dcv = torch.empty_like(du)
        dw = torch.empty_like(dv)
        ext().norm_project(du, ucv, inv_c, dcv)
        ext().norm_project(dv, uw, inv_w, dw)
        return dcv, dw, None, None, None, None


def angular_margin_head_hip(cv_bf16, w, label, cos_m, sin_m, s):
    """Gated entry: HIP kernels when the shape fits, torch reference
    otherwise (CPU, EP != 128)."""
    from . import reference as R

    if cv_bf16.is_cuda and cv_bf16.shape[1] == 128 and w.shape[1] == 128:
        return AngularMarginHead.apply(cv_bf16, w, label, cos_m, sin_m, s)
    return R.angular_margin_head(cv_bf16, w, label, cos_m, sin_m, s)


def head_logits_with_stats(cv_bf16, w, bias):
    """Forward logits for the FUSED head+loss path (no autograd tracking —
    gradients flow through FusedHeadLoss instead).  Uses the custom MFMA
    head_fwd (+ online-softmax stats epilogue) when gated on, else the
    hipBLASLt linear; either way the result carries what the fused loss
    needs stashed as tensor attributes."""
    B, EP = cv_bf16.shape
    L = w.shape[0]
    with torch.no_grad():
        if _HEAD_FWD and _HF_AIMG and EP == 128 and L % 8 == 0:
            # W pre-swizzled into the A-fragment image so the kernel's W
            # reads are contiguous 1-KB wave reads.  Originally the
            # large-L fix (strided per-lane row loads streamed W from
            # HBM inefficiently at L=261k), but measured faster than the
            # direct-load kernel at top11 scale too (63.7 vs 79.4 us at
            # L=72,416; step 1.337 vs 1.350 ms) — the swizzle cost
            # (~7 us, L2-resident) is well under the fragment-read win.
            wimg = _scratch_bf16("hf_aimg",
                                 ((L + 255) // 256 * 16, 4, 64, 8),
                                 w.device)
            ext().swizzle_a(w, wimg)
            logits = torch.empty(B, L, dtype=torch.bfloat16,
                                 device=cv_bf16.device)
            gx = (L + 255) // 256
            pm = torch.empty(gx, B, dtype=torch.float32, device=w.device)
            ps = torch.empty_like(pm)
            ext().head_fwd_img(cv_bf16, wimg, bias.float(), logits, pm, ps,
                               L)
            logits._c2v_lsm_partials = (pm, ps)
            logits._c2v_wimg_a = wimg  # reusable by the streaming backward
        elif _HEAD_FWD and EP % 32 == 0 and L <= _HEAD_FWD_MAXL:
            # direct-load kernel: the EP != 128 path (C2V_HF_AIMG=0 also
            # lands here below C2V_HEAD_FWD_MAXL)
            logits = torch.empty(B, L, dtype=torch.bfloat16,
                                 device=cv_bf16.device)
            gx = (L + 255) // 256
            pm = torch.empty(gx, B, dtype=torch.float32, device=w.device)
            ps = torch.empty_like(pm)
            ext().head_fwd(cv_bf16, w, bias.float(), logits, pm, ps)
            logits._c2v_lsm_partials = (pm, ps)
        else:
            logits = torch.nn.functional.linear(
                cv_bf16, w, bias.to(torch.bfloat16))
    return logits


class FusedHeadLoss(torch.autograd.Function):
    """K10+K12 collapsed into one autograd node (recompute-G backward).

    Forward: merges the head epilogue's online-softmax partials into
    (lse, acc) — the [B, L] logits are never re-read in full when the
    custom head forward ran.  Backward: dW/dbias (head_bwd_dw) and dcv
    (head_bwd_dcv) recompute G[b,l] = coef_b*(exp(logit-lse_b) - [l==y_b])
    from logits+lse in registers; the dlogits tensor of the unfused chain
    (a 61 MB write + 3x 61 MB reads at top11, 534 MB x4 at java-large) is
    never materialized.  Reference math: main.py:251-264 composed with
    model/model.py:83.

    ``logits`` is a NON-differentiable input (computed under no_grad by
    head_logits_with_stats); gradients flow to cv/w/bias directly.
    """

    @staticmethod
    def forward(ctx, logits, cv_bf16, w, bias, label, weight):
        B, L = logits.shape
        lse = torch.empty(B, dtype=torch.float32, device=logits.device)
        partials = getattr(logits, "_c2v_lsm_partials", None)
        if partials is not None:
            pm, ps = partials
            del logits._c2v_lsm_partials
            acc = _finalize_stats(logits, pm, ps, label, weight, lse)
        else:
            acc = _lsm_stats(logits, label, weight, lse)
        loss = acc[0] / acc[1]
        ctx.save_for_backward(logits, cv_bf16, w, label, weight, lse, acc)
        # saved_tensors re-wraps the tensor object, so python attributes
        # do not survive — carry the forward's W fragment image on ctx
        ctx.wimg_a = getattr(logits, "_c2v_wimg_a", None)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        logits, cv, w, label, weight, lse, acc = ctx.saved_tensors
        B, L = logits.shape
        dev = w.device
        g = dloss.contiguous().float().view(1)
        # per-row (coef, lse, y) packed once: ONE 16-B load per staged
        # logits chunk in the kernels below
        coef_lse = _scratch_f32("head_coef_lse", (B, 4), dev)
        ext().head_bwd_prep(label, weight, acc, g, lse, coef_lse)
        # dW + dbias (label-major): cv pre-swizzled once (256 KB) into the
        # MFMA fragment-image layout so every B-fragment read in the dW
        # kernel is a contiguous 1-KB wave read from L2
        nchunk = (B + 63) // 64 * 2
        cvimg = _scratch_bf16("head_cvimg", (nchunk, 8, 64, 8), dev)
        ext().swizzle_cv(cv.contiguous(), cvimg)
        dw = torch.empty(L, 128, dtype=torch.bfloat16, device=dev)
        dbias = torch.empty(L, dtype=torch.float32, device=dev)
        ext().head_bwd_dw(logits, cvimg, coef_lse, dw, dbias)
        # dcv (batch-major split-K): label chunk balances split-K slab
        # traffic against grid width (A/B-swept: 1024 beats 512 by ~30
        # us/step at top11; 4096 at java-large keeps slabs ~33 MB).  W is
        # pre-swizzled into the B-fragment image the kernel stages from
        # (one contiguous 32-KB LDS copy per sub-stage; replaces the
        # transpose_w kernel + line-amplified transposed-W staging).
        chunk = int(os.environ.get("C2V_HB_DCV_CHUNK", "0")) or (
            1024 if L <= 131072 else 4096)
        split = (L + chunk - 1) // chunk
        wimg = _scratch_bf16("head_wimg", ((L + 127) // 128 * 4, 8, 64, 8),
                             dev)
        ext().swizzle_cv(w, wimg)
        partials = _scratch_f32("head_fused_dcv", (split, B, 128), dev)
        if _HB_RC:
            # STREAMING dcv: the [B, L] logits are never read — each wave
            # recomputes its logits tile by MFMA from the swizzle_a
            # fragment images of cv and W (the W image is reused from the
            # forward when head_fwd_img ran)
            cvimg_a = _scratch_bf16(
                "head_cvimg_a", ((B + 255) // 256 * 16, 4, 64, 8), dev)
            ext().swizzle_a(cv.contiguous(), cvimg_a)
            wimg_a = ctx.wimg_a
            if wimg_a is None:
                wimg_a = _scratch_bf16(
                    "hf_aimg", ((L + 255) // 256 * 16, 4, 64, 8), dev)
                ext().swizzle_a(w, wimg_a)
            ext().head_bwd_dcv_rc(cvimg_a, wimg_a, wimg, coef_lse,
                                  partials, B, L, chunk)
        else:
            ext().head_bwd_dcv(logits, wimg, coef_lse, partials, chunk)
        dcv = torch.empty(B, 128, dtype=torch.bfloat16, device=dev)
        ext().slab_sum_bf16(partials, dcv)
        return None, dcv, dw, dbias, None, None


def fused_head_loss_supported(cv_bf16, w, training: bool) -> bool:
    """Shape/mode gates for the fused head+loss path."""
    B, EP = cv_bf16.shape
    L = w.shape[0]
    return (FUSED_HEAD and training and torch.is_grad_enabled()
            and cv_bf16.is_cuda and EP == 128 and L % 8 == 0 and B % 8 == 0
            and L < (1 << 24))  # y carried as f32 in the G recompute


def row_max_argmax(logits: torch.Tensor):
    """K17: per-row (max, argmax) for the eval/export prediction path —
    the last eval op that went through a torch kernel.  Falls back to
    torch.max off the HIP bf16 path."""
    if logits.is_cuda and logits.dtype == torch.bfloat16 \
            and logits.is_contiguous():
        B = logits.shape[0]
        vals = torch.empty(B, dtype=torch.float32, device=logits.device)
        idx = torch.empty(B, dtype=torch.int64, device=logits.device)
        ext().row_max_argmax(logits, vals, idx)
        return vals, idx
    return torch.max(logits.float(), dim=1)


def adam_step(
    param: torch.Tensor,
    grad: torch.Tensor,
    master: Optional[torch.Tensor],
    m: torch.Tensor,
    v: torch.Tensor,
    bc_pow: torch.Tensor,
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
) -> None:
    """K16: fused Adam.  bf16 params carry an f32 master (updated in f32,
    rounded to bf16 param); f32 params update in place (master is None).
    ``bc_pow`` is the optimizer's device-resident (beta1^t, beta2^t)
    float64 pair (advanced by adam_tick once per step)."""
    if param.dtype == torch.bfloat16:
        assert master is not None
        ext().adam_step_bf16(
            param.view(-1), grad.view(-1), master, m, v,
            bc_pow, lr, beta1, beta2, eps, weight_decay,
        )
    else:
        ext().adam_step_f32(
            param.view(-1), grad.view(-1).float(), m, v,
            bc_pow, lr, beta1, beta2, eps, weight_decay,
        )
