// Fused output-head + loss backward (recompute-G): dW, dbias, dcv straight
// from (logits, lse) — the [B, L] softmax gradient ("dlogits") is NEVER
// materialized.  Replaces the unfused chain lsm_nll_bwd (61 MB write) +
// colsum (61 MB read) + rocBLAS dW GEMM (61 MB read) + head_dgrad (61 MB
// read) with two kernels that each read logits once and recompute
//   G[b,l] = coef_b * (exp(logit[b,l] - lse_b) - [l == y_b]),
//   coef_b = gscale * weight[y_b] / sum_b weight[y_b]
// in registers (reference math: main.py:251-264 chained through
// model/model.py:83; G is the gradient of the weighted-NLL log-softmax
// composed with the linear head).
//
// Kernel 1 (label-major): dW[L, EP] = G^T @ cv and dbias[L] = colsum(G).
//   Block = 64 labels x EP(=128), K-loop over ALL batch rows in 64-row
//   stages.  G tiles are computed at stage time from coalesced logits row
//   chunks and written TRANSPOSED into LDS MFMA A-fragment images (the
//   wgrad.hip padded-image recipe — G^T fragment k-runs walk the batch
//   dim, which is logits' slow axis).  cv is pre-transposed once
//   ([EP, B], 256 KB — L2-resident) so B-fragment k-runs are contiguous
//   global loads.  dbias falls out of the same pass: each thread keeps
//   8 label-column partial sums of its staged G values, LDS-reduced at
//   the end; every label belongs to exactly ONE block, so the store is
//   a plain f32 write (no atomics).
//
// Kernel 2 (batch-major): dcv[B, EP] = G @ W — head_dgrad.hip's proven
//   split-K structure with the A fragment recomputed from contiguous
//   logits row chunks instead of read from dlogits.  The label chunk per
//   block is a runtime parameter: 512 at L <= 64k (59 slabs at top11),
//   4096 at java-large scale so the fp32 partial-slab traffic stays ~33 MB
//   instead of 268 MB.

#include "common.h"

#define HB_LB 64    // labels per dW block
#define HB_ROWS 64  // batch rows staged per K-iteration
#define HB_NSTR (64 * 8 + 8)  // padded A-image stride (wgrad recipe)

__global__ __launch_bounds__(512) void head_bwd_dw_kernel(
    const bf16* __restrict__ logits, const bf16* __restrict__ cvt,
    const float* __restrict__ lse, const long* __restrict__ label,
    const float* __restrict__ weight, const float* __restrict__ acc_ws,
    const float* __restrict__ gscale, bf16* __restrict__ dw,
    float* __restrict__ dbias, long B, long L) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const long l0 = (long)blockIdx.x * HB_LB;
  // staging role: one bf16x8 logits chunk per thread per stage
  const int krow = threadIdx.x >> 3;        // 0..63: batch row within stage
  const int col8 = (threadIdx.x & 7) * 8;   // 0..56: label chunk start
  const float inv_ws = gscale[0] / acc_ws[1];

  // A-fragment images (double-buffered) reused as the dbias reduction
  // scratch after the K-loop (lives separated by barriers)
  __shared__ union {
    bf16 img[2][2][4][HB_NSTR];  // [buf][ksub][label tile][fragment image]
    float red[HB_ROWS][HB_LB + 1];
  } sm;
  __shared__ float red2[8][HB_LB + 1];

  const int n = col8 >> 4;          // this thread's label tile
  const int base_l = col8 & 15;     // label offset within the tile
  const int ksub = krow >> 5;       // which 32-k MFMA step of the stage
  const int kgrp = (krow >> 3) & 3; // 16-lane fragment group
  const int jslot = krow & 7;

  float db[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) db[j] = 0.f;

  bf16 gv[8];
  auto stage_load = [&](long mb) {
    const long row = mb + krow;
    const long lc = l0 + col8;
    if (row < B && lc < L) {  // L % 8 == 0: chunk fully in bounds
      const long y = label[row];
      const float coef = inv_ws * (weight ? weight[y] : 1.f);
      const float lseb = lse[row];
      bf16 v[8];
      *(uint4*)v = *(const uint4*)(logits + row * L + lc);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = coef * __expf(bf2f(v[j]) - lseb);
        if (lc + j == y) g -= coef;
        const bf16 gb = f2bf(g);
        gv[j] = gb;
        db[j] += bf2f(gb);  // dbias from the rounded value: bitwise parity
                            // with colsum over a stored bf16 dlogits
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) gv[j] = f2bf(0.f);
    }
  };
  auto stage_write = [&](int buf) {
    bf16* dst = sm.img[buf][ksub][n];
#pragma unroll
    for (int j = 0; j < 8; ++j)
      dst[(base_l + j + kgrp * 16) * 8 + jslot] = gv[j];
  };

  f32x4 acc[4];
#pragma unroll
  for (int nt = 0; nt < 4; ++nt) acc[nt] = f32x4{0.f, 0.f, 0.f, 0.f};
  const bf16x8 zero8 = {};

  const int lt = wave >> 1;          // this wave's label tile
  const int nh = (wave & 1) * 64;    // this wave's EP half

  stage_load(0);
  stage_write(0);
  __syncthreads();
  int buf = 0;
  for (long mb = 0; mb < B; mb += HB_ROWS) {
    if (mb + HB_ROWS < B) stage_load(mb + HB_ROWS);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const bf16x8 a = *(const bf16x8*)&sm.img[buf][ks][lt][lane * 8];
      const long k0 = mb + ks * 32 + (lane >> 4) * 8;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const long colv = nh + nt * 16 + (lane & 15);
        const bf16x8 b =
            (k0 < B) ? *(const bf16x8*)(cvt + colv * B + k0) : zero8;
        acc[nt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt], 0, 0, 0);
      }
    }
    if (mb + HB_ROWS < B) stage_write(buf ^ 1);
    buf ^= 1;
    __syncthreads();
  }

  // dW: each wave owns [16 labels x 64 EP] of the block's [64, 128] tile
#pragma unroll
  for (int nt = 0; nt < 4; ++nt) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long lrow = l0 + lt * 16 + (lane >> 4) * 4 + r;
      if (lrow < L)
        dw[lrow * 128 + nh + nt * 16 + (lane & 15)] = f2bf(acc[nt][r]);
    }
  }

  // dbias: reduce the per-thread 8-label partials.  Write layout
  // red[krow][label]: bank-conflict-free both ways (stride 65).
#pragma unroll
  for (int j = 0; j < 8; ++j) sm.red[krow][col8 + j] = db[j];
  __syncthreads();
  {
    const int lbl = threadIdx.x & 63;
    const int kg0 = (threadIdx.x >> 6) * 8;
    float p = 0.f;
#pragma unroll
    for (int kg = 0; kg < 8; ++kg) p += sm.red[kg0 + kg][lbl];
    red2[threadIdx.x >> 6][lbl] = p;
  }
  __syncthreads();
  if (threadIdx.x < 64 && l0 + threadIdx.x < L) {
    float s = 0.f;
#pragma unroll
    for (int q = 0; q < 8; ++q) s += red2[q][threadIdx.x];
    dbias[l0 + threadIdx.x] = s;
  }
}

// dcv split-K partials: head_dgrad.hip's kernel with A = G recomputed.
__global__ __launch_bounds__(512) void head_bwd_dcv_kernel(
    const bf16* __restrict__ logits, const bf16* __restrict__ wt,
    const float* __restrict__ lse, const long* __restrict__ label,
    const float* __restrict__ weight, const float* __restrict__ acc_ws,
    const float* __restrict__ gscale, float* __restrict__ partials, long B,
    long L, int chunk, int GYB) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int total = gridDim.x;
  const int lin = (total % 8 == 0)
      ? (int)(blockIdx.x % 8) * (total / 8) + (int)blockIdx.x / 8
      : (int)blockIdx.x;
  const int by = lin % GYB;
  const int sc = lin / GYB;
  const long bt0 = (long)by * 128;
  const int kj = (lane >> 4) * 8;

  const long row = bt0 + wave * 16 + (lane & 15);
  const bool aok = row < B;
  float coef = 0.f, lseb = 0.f;
  long y = -1;
  if (aok) {
    y = label[row];
    coef = gscale[0] * (weight ? weight[y] : 1.f) / acc_ws[1];
    lseb = lse[row];
  }
  const bf16* ap = logits + row * L + kj;

  __shared__ bf16 wst[128][132];  // padded: conflict-free b128 reads

  f32x4 acc[8];
#pragma unroll
  for (int nt = 0; nt < 8; ++nt) acc[nt] = f32x4{0.f, 0.f, 0.f, 0.f};
  const bf16x8 zero8 = {};

  const int nsub = chunk / 128;
  for (int s = 0; s < nsub; ++s) {
    const long l0 = (long)sc * chunk + (long)s * 128;
    if (l0 >= L) break;
    for (int t = threadIdx.x; t < 128 * 16; t += 512) {
      const int e = t >> 4;
      const int c = t & 15;
      const bf16x8 v = (l0 + c * 8 < L)
          ? *(const bf16x8*)(wt + (long)e * L + l0 + c * 8) : zero8;
      *(bf16x8*)&wst[e][c * 8] = v;
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      const long lc = l0 + kk * 32 + kj;
      bf16x8 a = zero8;
      if (aok && lc < L) {
        bf16 v[8];
        *(uint4*)v = *(const uint4*)(ap + l0 + kk * 32);
        bf16 g[8];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          float gg = coef * __expf(bf2f(v[j]) - lseb);
          if (lc + j == y) gg -= coef;
          g[j] = f2bf(gg);
        }
        a = *(const bf16x8*)g;
      }
#pragma unroll
      for (int nt = 0; nt < 8; ++nt) {
        const bf16x8 b =
            *(const bf16x8*)&wst[nt * 16 + (lane & 15)][kk * 32 + kj];
        acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt],
                                                          0, 0, 0);
      }
    }
    __syncthreads();
  }

  float* slab = partials + (long)sc * B * 128;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const long grow = bt0 + wave * 16 + (lane >> 4) * 4 + r;
    if (grow < B) {
#pragma unroll
      for (int nt = 0; nt < 8; ++nt)
        slab[grow * 128 + nt * 16 + (lane & 15)] = acc[nt][r];
    }
  }
}

extern "C" {

void launch_head_bwd_dw(const void* logits, const void* cvt, const float* lse,
                        const long* label, const float* weight,
                        const float* acc_ws, const float* gscale, void* dw,
                        float* dbias, long B, long L, hipStream_t stream) {
  const int grid = (int)((L + HB_LB - 1) / HB_LB);
  head_bwd_dw_kernel<<<grid, 512, 0, stream>>>(
      (const bf16*)logits, (const bf16*)cvt, lse, label, weight, acc_ws,
      gscale, (bf16*)dw, dbias, B, L);
}

void launch_head_bwd_dcv(const void* logits, const void* wt, const float* lse,
                         const long* label, const float* weight,
                         const float* acc_ws, const float* gscale,
                         float* partials, long B, long L, int chunk,
                         hipStream_t stream) {
  const int GYB = (int)((B + 127) / 128);
  const int split = (int)((L + chunk - 1) / chunk);
  head_bwd_dcv_kernel<<<GYB * split, 512, 0, stream>>>(
      (const bf16*)logits, (const bf16*)wt, lse, label, weight, acc_ws,
      gscale, partials, B, L, chunk, GYB);
}

}  // extern "C"
