"""Localize head_fwd mismatches: per-tile relative error map + samples."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from code2vec_amd.ops import ext  # noqa: E402


def main():
    dev = torch.device("cuda:0")
    for (B, L, EP) in [(24, 700, 128), (64, 512, 128), (1024, 30000, 128)]:
        g = torch.Generator().manual_seed(11)
        cv = (torch.randn(B, EP, generator=g) * 0.5).to(dev, torch.bfloat16)
        w = (torch.randn(L, EP, generator=g) * 0.1).to(dev, torch.bfloat16)
        bias = torch.randn(L, generator=g).to(dev)
        out = torch.full((B, L), float("nan"), dtype=torch.bfloat16,
                         device=dev)
        gx = (L + 255) // 256
        pm = torch.empty(gx, B, dtype=torch.float32, device=dev)
        ps = torch.empty_like(pm)
        ext().head_fwd(cv, w, bias, out, pm, ps)
        torch.cuda.synchronize()
        ref = cv.float() @ w.float().t() + bias
        o = out.float()
        err = (o - ref).abs()
        print(f"B={B} L={L}: nan={torch.isnan(o).sum().item()} "
              f"maxerr={err.nan_to_num(1e9).max().item():.4f} "
              f"relerr={(err.nan_to_num(1e9).norm()/ref.norm()).item():.4f}")
        # per 16x16 tile error map (coarse): which tiles are wrong?
        bt, lt = min(B, 64), min(L, 256)
        em = err[:bt, :lt].nan_to_num(1e9).view(bt // 8, 8, lt // 16, 16)
        tile = em.amax(dim=(1, 3))
        bad = (tile > 0.1).nonzero()
        print("  bad 8x16 tiles (row-blk, col-blk):", bad[:12].tolist(),
              f"of {tile.numel()}")
        i = (err.nan_to_num(1e9) > 0.1).nonzero()
        if len(i):
            r, c = i[0].tolist()
            print(f"  sample out[{r},{c}]={o[r,c].item():.4f} "
                  f"ref={ref[r,c].item():.4f}")
            r, c = i[-1].tolist()
            print(f"  last bad [{r},{c}] out={o[r,c].item():.4f} "
                  f"ref={ref[r,c].item():.4f}  n_bad={len(i)}")


if __name__ == "__main__":
    main()
