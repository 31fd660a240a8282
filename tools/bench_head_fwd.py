"""Microbench: custom head_fwd (with/without fused loss stats) vs
hipBLASLt linear + full-pass lsm forward, at the top11 shape.

Run on a GPU box:  python tools/bench_head_fwd.py
"""
import os
import sys

os.environ.setdefault("PYTORCH_TUNABLEOP_ENABLED", "1")
os.environ.setdefault("PYTORCH_TUNABLEOP_TUNING", "1")

import torch  # noqa: E402

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from code2vec_amd.ops import ext  # noqa: E402


def time_op(fn, iters=200, warmup=20):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(iters):
        fn()
    e.record()
    torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000  # us


def main():
    dev = torch.device("cuda:0")
    B, L, EP = 1024, 30000, 128
    g = torch.Generator().manual_seed(0)
    cv = (torch.randn(B, EP, generator=g) * 0.5).to(dev, torch.bfloat16)
    w = (torch.randn(L, EP, generator=g) * 0.1).to(dev, torch.bfloat16)
    bias = torch.randn(L, generator=g).to(dev)
    bias_bf = bias.to(torch.bfloat16)
    out = torch.empty(B, L, dtype=torch.bfloat16, device=dev)
    gx = (L + 255) // 256
    pm = torch.empty(gx, B, dtype=torch.float32, device=dev)
    ps = torch.empty_like(pm)
    none = torch.Tensor()
    label = torch.randint(0, L, (B,), generator=g).to(dev)
    weight = torch.ones(L, device=dev)
    lse = torch.empty(B, dtype=torch.float32, device=dev)
    acc = torch.zeros(2, dtype=torch.float32, device=dev)

    if os.environ.get("C2V_HF_VARIANT"):
        t = time_op(lambda: ext().head_fwd(cv, w, bias, out, none, none))
        print(f"variant {os.environ['C2V_HF_VARIANT']}: {t:7.1f} us")
        return

    t_lin = time_op(lambda: torch.nn.functional.linear(cv, w, bias_bf))
    t_hf = time_op(lambda: ext().head_fwd(cv, w, bias, out, none, none))
    t_hfs = time_op(lambda: ext().head_fwd(cv, w, bias, out, pm, ps))
    t_lsm = time_op(lambda: ext().logsoftmax_nll_fwd(out, label, weight,
                                                     lse, acc))
    t_fin = time_op(lambda: ext().logsoftmax_nll_finalize(out, pm, ps, label,
                                                          weight, lse, acc))
    # head backward dcv: library GEMM vs split-K custom
    t_dcv_lib = time_op(lambda: out @ w)
    wt = w.t().contiguous()
    split = (L + 511) // 512
    partials = torch.empty(split, B, 128, dtype=torch.float32, device=dev)
    dcv_out = torch.empty(B, 128, dtype=torch.bfloat16, device=dev)
    t_tr = time_op(lambda: ext().transpose_w(w, wt))
    t_dcv = time_op(lambda: ext().head_dgrad(out, wt, partials))
    t_red = time_op(lambda: ext().slab_sum_bf16(partials, dcv_out))
    print(f"dcv rocBLAS:                  {t_dcv_lib:7.1f} us")
    print(f"dcv custom (tr+k+reduce):     {t_tr + t_dcv + t_red:7.1f} us "
          f"({t_tr:.1f} + {t_dcv:.1f} + {t_red:.1f})")

    bw = (B * L * 2) / 1e9
    print(f"linear (hipBLASLt+TunableOp): {t_lin:7.1f} us")
    print(f"head_fwd no-stats:            {t_hf:7.1f} us "
          f"({bw / (t_hf / 1e6):.2f} GB/s wr)")
    print(f"head_fwd with-stats:          {t_hfs:7.1f} us")
    print(f"lsm_nll_fwd full pass:        {t_lsm:7.1f} us")
    print(f"lsm_finalize:                 {t_fin:7.1f} us")
    print(f"old path (linear+lsm):        {t_lin + t_lsm:7.1f} us")
    print(f"new path (fused+finalize):    {t_hfs + t_fin:7.1f} us")


if __name__ == "__main__":
    main()
