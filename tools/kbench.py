#!/usr/bin/env python3
"""Per-op GPU micro-benchmarks (within-process A/B, CUDA events).

Times each framework op at the top11 flagship shape; variants switch via
env vars read by ops/functional.py.  Usage (on a GPU box):
    python tools/kbench.py [op ...]
"""

from __future__ import annotations

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from code2vec_amd.ops import functional as Fn
from code2vec_amd.ops import round_up


def timeit(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters * 1000  # us


def main():
    if "--help" in sys.argv or "-h" in sys.argv:
        print(__doc__.strip())
        return
    if not torch.cuda.is_available():
        print("kbench: needs a GPU (per-kernel timing); run under gpurun")
        return
    dev = torch.device("cuda:0")
    B, C = 1024, 200
    T, P, L = 360632, 342846, 30000
    L = int(os.environ.get("C2V_KBENCH_L", L))  # e.g. 261000 (java-large)
    T = int(os.environ.get("C2V_KBENCH_T", T))
    dt = dp = E = 100
    TS, PS, EP = round_up(dt), round_up(dp), round_up(E)
    KP = 2 * TS + PS
    M = B * C
    g = torch.Generator(device=dev).manual_seed(0)

    term = torch.randn(T, TS, generator=g, device=dev).to(torch.bfloat16)
    path = torch.randn(P, PS, generator=g, device=dev).to(torch.bfloat16)
    starts = torch.randint(1, T, (B, C), generator=g, device=dev, dtype=torch.int32)
    pth = torch.randint(1, P, (B, C), generator=g, device=dev, dtype=torch.int32)
    ends = torch.randint(1, T, (B, C), generator=g, device=dev, dtype=torch.int32)
    x = Fn.GatherConcat.apply(starts, pth, ends, term, path)
    w = (torch.randn(EP, KP, generator=g, device=dev) * 0.05).to(torch.bfloat16)
    gamma = torch.rand(EP, generator=g, device=dev) + 0.5
    beta = torch.randn(EP, generator=g, device=dev) * 0.1
    a = torch.randn(EP, generator=g, device=dev) * 0.2
    wout = (torch.randn(L, EP, generator=g, device=dev) * 0.05).to(torch.bfloat16)
    label = torch.randint(0, L, (B,), generator=g, device=dev)
    weight = torch.ones(L, device=dev)

    ops = sys.argv[1:] or ["gather_fwd", "gather_bwd", "combiner_fwd",
                           "combiner_bwd", "attention", "lsm", "adam",
                           "wgrad"]
    results = {}

    if "gather_fwd" in ops:
        results["gather_fwd"] = timeit(
            lambda: Fn.GatherConcat.apply(starts, pth, ends, term, path))

    if "gather_bwd" in ops:
        xg = None
        def run():
            nonlocal xg
            t2 = term.detach().requires_grad_(True)
            p2 = path.detach().requires_grad_(True)
            o = Fn.GatherConcat.apply(starts, pth, ends, t2, p2)
            o.backward(x)
        results["gather_bwd(all)"] = timeit(run, iters=15)

    if "combiner_fwd" in ops:
        results["combiner_fwd"] = timeit(
            lambda: Fn.CombinerLNTanh.apply(x, w, gamma, beta, E, 0.25, True))

    if "combiner_bwd" in ops:
        def run():
            x2 = x.detach().requires_grad_(True)
            w2 = w.detach().requires_grad_(True)
            o = Fn.CombinerLNTanh.apply(x2, w2, gamma, beta, E, 0.0, False)
            o.backward(o.detach())
        results["combiner_fwd+bwd(all)"] = timeit(run, iters=15)

    ccv = Fn.CombinerLNTanh.apply(x, w, gamma, beta, E, 0.0, False).view(B, C, EP)
    if "attention" in ops:
        results["attention_fwd"] = timeit(
            lambda: Fn.AttentionPool.apply(ccv, a, starts, E))

    if "lsm" in ops:
        cv, _ = Fn.AttentionPool.apply(ccv, a, starts, E)
        logits = (cv.to(torch.bfloat16) @ wout.t()).contiguous()
        results["lsm_fwd"] = timeit(
            lambda: Fn.FusedLogSoftmaxNLL.apply(logits, label, weight))

    if "membw" in ops:
        # same-box streaming ceiling: D2D copy (read+write) of 2 GB
        src = torch.empty(1 << 28, dtype=torch.float32, device=dev)
        dst = torch.empty_like(src)
        src.fill_(1.0)
        t = timeit(lambda: dst.copy_(src), iters=10)
        results["d2d_copy(2GB r+w)"] = t
        print(f"# copy bandwidth: {2 * src.numel() * 4 / (t * 1e-6) / 1e12:.2f} TB/s")

    if "adam" in ops:
        p1 = term.clone().view(-1)
        g1 = term.clone().view(-1)
        master = p1.float()
        m = torch.zeros_like(master)
        v = torch.zeros_like(master)
        from code2vec_amd.ops import ext
        bc = torch.full((2,), 0.5, dtype=torch.float64, device=dev)
        results["adam(term)"] = timeit(
            lambda: ext().adam_step_bf16(p1, g1, master, m, v, bc, 0.01,
                                         0.9, 0.999, 1e-8, 0.0))

    if "wgrad" in ops:
        dz = ccv.view(M, EP).contiguous()
        partials = torch.empty(256, KP, EP, dtype=torch.float32, device=dev)
        from code2vec_amd.ops import ext
        results["wgrad"] = timeit(lambda: ext().wgrad(x, dz, partials))

    if "head_fwd" in ops:
        from code2vec_amd.ops import ext
        cv, _ = Fn.AttentionPool.apply(ccv, a, starts, E)
        cvb = cv.to(torch.bfloat16).contiguous()
        bias = torch.zeros(L, device=dev)
        logits = torch.empty(B, L, dtype=torch.bfloat16, device=dev)
        gx = (L + 255) // 256
        pm = torch.empty(gx, B, dtype=torch.float32, device=dev)
        ps = torch.empty_like(pm)
        nil = torch.Tensor()
        results["head_fwd(no stats)"] = timeit(
            lambda: ext().head_fwd(cvb, wout, bias, logits, nil, nil),
            iters=10)
        results["head_fwd(stats)"] = timeit(
            lambda: ext().head_fwd(cvb, wout, bias, logits, pm, ps),
            iters=10)
        lse = torch.empty(B, dtype=torch.float32, device=dev)
        acc_f = torch.empty((B + 15) // 16, 2, dtype=torch.float32, device=dev)
        results["lsm_finalize"] = timeit(
            lambda: ext().logsoftmax_nll_finalize(logits, pm, ps, label,
                                                  weight, lse, acc_f),
            iters=10)
        gx2 = (L + 16383) // 16384
        pm2 = torch.empty(gx2, B, dtype=torch.float32, device=dev)
        ps2 = torch.empty_like(pm2)
        results["lsm_partial+final"] = timeit(
            lambda: (ext().lsm_partial(logits, pm2, ps2),
                     ext().logsoftmax_nll_finalize(logits, pm2, ps2, label,
                                                   weight, lse, acc_f)),
            iters=10)
        acc_b = torch.empty(B, 2, dtype=torch.float32, device=dev)
        results["lsm_full_fwd"] = timeit(
            lambda: ext().logsoftmax_nll_fwd(logits, label, weight, lse,
                                             acc_b), iters=10)
        bias_h = bias.to(torch.bfloat16)
        results["linear(hipblaslt)"] = timeit(
            lambda: torch.nn.functional.linear(cvb, wout, bias_h), iters=10)
        wimg_a = torch.empty((L + 255) // 256 * 16, 4, 64, 8,
                             dtype=torch.bfloat16, device=dev)
        ext().swizzle_a(wout, wimg_a)
        results["swizzle_a"] = timeit(lambda: ext().swizzle_a(wout, wimg_a),
                                      iters=10)
        results["head_fwd_img(stats)"] = timeit(
            lambda: ext().head_fwd_img(cvb, wimg_a, bias, logits, pm, ps,
                                       L), iters=10)

    if "head_bwd" in ops:
        from code2vec_amd.ops import ext
        cv, _ = Fn.AttentionPool.apply(ccv, a, starts, E)
        cvb = cv.to(torch.bfloat16).contiguous()
        bias = torch.zeros(L, device=dev)
        logits = torch.empty(B, L, dtype=torch.bfloat16, device=dev)
        gx = (L + 255) // 256
        pm = torch.empty(gx, B, dtype=torch.float32, device=dev)
        ps = torch.empty_like(pm)
        ext().head_fwd(cvb, wout, bias, logits, pm, ps)
        lse = torch.empty(B, dtype=torch.float32, device=dev)
        acc_f = torch.empty((B + 15) // 16, 2, dtype=torch.float32, device=dev)
        ext().logsoftmax_nll_finalize(logits, pm, ps, label, weight, lse, acc_f)
        acc = torch.empty(2, dtype=torch.float32, device=dev)
        ext().slab_sum_f32(acc_f, acc)
        g1 = torch.ones(1, device=dev)
        coef_lse = torch.zeros(B, 4, device=dev)
        ext().head_bwd_prep(label, weight, acc, g1, lse, coef_lse)
        nchunk = (B + 63) // 64 * 2
        cvimg = torch.empty(nchunk, 8, 64, 8, dtype=torch.bfloat16, device=dev)
        ext().swizzle_cv(cvb, cvimg)
        dw = torch.empty(L, 128, dtype=torch.bfloat16, device=dev)
        dbias = torch.empty(L, dtype=torch.float32, device=dev)
        results["swizzle_cv"] = timeit(lambda: ext().swizzle_cv(cvb, cvimg))
        results["head_bwd_dw"] = timeit(
            lambda: ext().head_bwd_dw(logits, cvimg, coef_lse, dw, dbias))
        chunk = int(os.environ.get("C2V_HB_DCV_CHUNK", "0")) or (
            1024 if L <= 131072 else 4096)
        split = (L + chunk - 1) // chunk
        partials = torch.empty(split, B, 128, dtype=torch.float32, device=dev)
        wimg = torch.empty((L + 127) // 128 * 4, 8, 64, 8,
                           dtype=torch.bfloat16, device=dev)
        ext().swizzle_cv(wout, wimg)
        results["swizzle_w"] = timeit(lambda: ext().swizzle_cv(wout, wimg))
        results["head_bwd_dcv"] = timeit(
            lambda: ext().head_bwd_dcv(logits, wimg, coef_lse, partials,
                                       chunk))
        wimg_a = torch.empty((L + 255) // 256 * 16, 4, 64, 8,
                             dtype=torch.bfloat16, device=dev)
        ext().swizzle_a(wout, wimg_a)
        cvimg_a = torch.empty((B + 255) // 256 * 16, 4, 64, 8,
                              dtype=torch.bfloat16, device=dev)
        ext().swizzle_a(cvb, cvimg_a)
        results["swizzle_a(cv)"] = timeit(
            lambda: ext().swizzle_a(cvb, cvimg_a))
        results["head_bwd_dcv_rc"] = timeit(
            lambda: ext().head_bwd_dcv_rc(cvimg_a, wimg_a, wimg, coef_lse,
                                          partials, B, L, chunk))

    for k, vv in results.items():
        print(f"{k:24s} {vv:10.1f} us")


if __name__ == "__main__":
    main()
