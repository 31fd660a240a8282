#!/usr/bin/env python3
"""Minimal-dispatch probe for PMC counter collection (rocprofv3 --pmc ...).

Runs each hot kernel a few times at the top11 flagship shape with as few
auxiliary dispatches as possible (PMC collection serializes and replays
every dispatch, so bench.py under --pmc is prohibitively slow).
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from code2vec_amd.ops import ext
from code2vec_amd.ops import functional as Fn
from code2vec_amd.ops import round_up

dev = torch.device("cuda:0")
B, C = 1024, 200
T, P = 360632, 342846
dt = dp = E = 100
TS, PS, EP = round_up(dt), round_up(dp), round_up(E)
KP = 2 * TS + PS
M = B * C
REPS = 4

g = torch.Generator(device=dev).manual_seed(0)
term = torch.randn(T, TS, generator=g, device=dev).to(torch.bfloat16)
path = torch.randn(P, PS, generator=g, device=dev).to(torch.bfloat16)
starts = torch.randint(1, T, (B, C), generator=g, device=dev, dtype=torch.int32)
pth = torch.randint(1, P, (B, C), generator=g, device=dev, dtype=torch.int32)
ends = torch.randint(1, T, (B, C), generator=g, device=dev, dtype=torch.int32)
w = (torch.randn(EP, KP, generator=g, device=dev) * 0.05).to(torch.bfloat16)
gamma = torch.rand(EP, generator=g, device=dev) + 0.5
beta = torch.randn(EP, generator=g, device=dev) * 0.1
a = torch.randn(EP, generator=g, device=dev) * 0.2

x = torch.empty(M, KP, dtype=torch.bfloat16, device=dev)
out = torch.empty(M, EP, dtype=torch.bfloat16, device=dev)
z = torch.empty(M, EP, dtype=torch.bfloat16, device=dev)
mean = torch.empty(M, dtype=torch.float32, device=dev)
rstd = torch.empty(M, dtype=torch.float32, device=dev)
cv = torch.empty(B, EP, dtype=torch.float32, device=dev)
attn = torch.empty(B, C, dtype=torch.float32, device=dev)
partials = torch.empty(256, KP, EP, dtype=torch.float32, device=dev)

for _ in range(REPS):
    ext().gather_concat_fwd(starts, pth, ends, term, path, x)
for _ in range(REPS):
    ext().combiner_fwd(x, w, gamma, beta, out, z, mean, rstd, E, 0.0, 0,
                       torch.zeros(1, dtype=torch.int64, device=dev), 1)
ccv = out.view(B, C, EP)
for _ in range(REPS):
    ext().attention_fwd(ccv, a, starts, cv, attn, E)
for _ in range(REPS):
    ext().wgrad(x, out, partials)

# scatter path (pre-grouped outside the timed kernels)
idx_se = torch.cat([starts.view(-1), ends.view(-1)])
sorted_se, perm_se, counts_se = Fn._group_by_index(idx_se, T)
dterm32 = torch.zeros(T, TS, dtype=torch.float32, device=dev)
dterm = torch.empty(T, TS, dtype=torch.bfloat16, device=dev)
flags_t = torch.zeros(T, dtype=torch.uint8, device=dev)
for _ in range(REPS):
    ext().embed_scatter_sorted(sorted_se, perm_se, x, dterm32, dterm,
                               flags_t, M, KP, 0, TS + PS, 16)
for _ in range(REPS):
    ext().cast_clear_rows(dterm32, counts_se, flags_t, dterm)

# head forward + fused backward kernels at the derived top11 vocab
L = 72416
wout = (torch.randn(L, EP, generator=g, device=dev) * 0.05).to(torch.bfloat16)
label = torch.randint(0, L, (B,), generator=g, device=dev)
weight = torch.ones(L, device=dev)
bias = torch.zeros(L, device=dev)
logits = torch.empty(B, L, dtype=torch.bfloat16, device=dev)
gx = (L + 255) // 256
pm = torch.empty(gx, B, dtype=torch.float32, device=dev)
ps = torch.empty_like(pm)
for _ in range(REPS):
    ext().head_fwd(cv.to(torch.bfloat16), wout, bias, logits, pm, ps)
wimg_a = torch.empty((L + 255) // 256 * 16, 4, 64, 8, dtype=torch.bfloat16,
                     device=dev)
ext().swizzle_a(wout, wimg_a)
for _ in range(REPS):
    ext().head_fwd_img(cv.to(torch.bfloat16), wimg_a, bias, logits, pm, ps,
                       L)
lse = torch.empty(B, dtype=torch.float32, device=dev)
acc_f = torch.empty((B + 15) // 16, 2, dtype=torch.float32, device=dev)
ext().logsoftmax_nll_finalize(logits, pm, ps, label, weight, lse, acc_f)
acc = torch.empty(2, dtype=torch.float32, device=dev)
ext().slab_sum_f32(acc_f, acc)
coef_lse = torch.zeros(B, 4, device=dev)
g1s = torch.ones(1, device=dev)
ext().head_bwd_prep(label, weight, acc, g1s, lse, coef_lse)
cvb = cv.to(torch.bfloat16).contiguous()
cvimg = torch.empty((B + 63) // 64 * 2, 8, 64, 8, dtype=torch.bfloat16,
                    device=dev)
ext().swizzle_cv(cvb, cvimg)
dwh = torch.empty(L, 128, dtype=torch.bfloat16, device=dev)
dbias = torch.empty(L, dtype=torch.float32, device=dev)
for _ in range(REPS):
    ext().head_bwd_dw(logits, cvimg, coef_lse, dwh, dbias)
wimg = torch.empty((L + 127) // 128 * 4, 8, 64, 8, dtype=torch.bfloat16,
                   device=dev)
ext().swizzle_cv(wout, wimg)
chunk = 1024
dcv_p = torch.empty((L + chunk - 1) // chunk, B, 128, dtype=torch.float32,
                    device=dev)
for _ in range(REPS):
    ext().head_bwd_dcv(logits, wimg, coef_lse, dcv_p, chunk)

p1 = term.view(-1)
g1 = term.clone().view(-1)
master = p1.float()
m_ = torch.zeros_like(master)
v_ = torch.zeros_like(master)
for _ in range(REPS):
    ext().adam_step_bf16(p1, g1, master, m_, v_,
                         torch.full((2,), 0.5, dtype=torch.float64,
                                    device=dev), 0.01, 0.9, 0.999, 1e-8,
                         0.0)

torch.cuda.synchronize()
print("pmc probe done")
