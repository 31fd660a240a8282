// Output-head forward: logits[B, L] = cv[B, EP] @ W[L, EP]^T + bias[L],
// optionally fused with the per-row online-softmax statistics that the
// log-softmax/NLL loss needs (reference main.py:251-264).
//
// Why custom: hipBLASLt treats this as an ordinary GEMM, but with K = EP
// (128) it is really a streaming op — 61 MB of C writes against 8 GFLOP.
// Computed TRANSPOSED (labels = MFMA rows, batch = MFMA cols): the C
// fragment then gives every lane 4 CONSECUTIVE label outputs, which pack
// into one 8-B nontemporal store — no LDS bounce, no barrier, no scalar
// stores.  (The row-major orientation needs an LDS transpose whose
// 2-B scalar writes are 8-way bank-conflicted and dominate the kernel —
// measured 61 us vs 23 us for this one; see PERF.md.)  Both fragments
// load straight from global memory as contiguous 16-B runs: the
// A-fragment k-run is a row of W[L, EP], the B-fragment k-run is a row
// of cv.  Fusing the loss statistics into the epilogue then deletes the
// loss kernel's full re-read of logits (61 MB at L = 30k).
//
// Block = 512 threads (8 waves), tile = 256 labels x 64 batch; wave w
// owns labels w*32..w*32+31 (2 MFMA row-tiles) x all 4 col-tiles.
// Batch-block index is INNER in blockIdx so the 16 concurrent blocks of
// one label tile share its W rows in L2 (W streams from HBM once).
// Per-batch-row (max, sumexp) partials over each 256-label block land in
// pm/ps[lab_block][b]; lsm_finalize_kernel merges them into lse + the
// weighted-NLL accumulators exactly as lsm_nll_fwd_kernel's tail does
// (logsoftmax_nll.hip).

#include <cstdlib>

#include "common.h"

#define HF_LABS 256  // labels per block
#define HF_BATCH 64  // batch cols per block
#define NINF (-3.0e38f)

// merge (m2, s2) into (m, s): online-softmax pair combine
#define HF_MERGE(m, s, m2, s2)                       \
  do {                                               \
    const float nm_ = fmaxf(m, m2);                  \
    s = s * __expf(m - nm_) + s2 * __expf(m2 - nm_); \
    m = nm_;                                         \
  } while (0)

// variant: phase-isolation bitmask for perf experiments (C2V_HF_VARIANT):
//   1 = skip global stores, 4 = skip B (cv) loads, 8 = skip A (w) loads,
//   16 = skip the LDS transpose writes
// AIMG=1: the A operand arrives as the swizzle_a fragment image (wimg
// aliasing the w pointer) — contiguous 1-KB wave reads instead of
// 256-B-strided per-lane row loads (the large-L fix; EP=128 only).
// MINW=6: the unconstrained build allocates 92 VGPR (2 resident blocks =
// 4 waves/SIMD); a forced 6-wave cap repacks to 74 VGPR with ZERO spill,
// and LDS (37.9 KB) still fits the 3rd block.  C2V_HF_OCC=1 reverts.
template <int NKT, int STATS, int AIMG = 0, int MINW = 6>
__global__ __launch_bounds__(512, MINW) void head_fwd_kernel(
    const bf16* __restrict__ cv, const bf16* __restrict__ w,
    const float* __restrict__ bias, bf16* __restrict__ out,
    float* __restrict__ pm, float* __restrict__ ps, long B, long L, int EP,
    int GYB, int variant) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  // XCD-aware swizzle: the dispatcher assigns block n to XCD n%8, which
  // would spread the GYB sibling blocks of one W tile (same bx) over all
  // 8 XCD L2s — W would stream from HBM once PER XCD.  Remap so each
  // XCD owns a contiguous bx range and the tile is pulled from HBM once.
  // (bijective only when the grid divides evenly; small/odd grids keep
  // the identity mapping)
  const int total = gridDim.x;
  const int lin = (total % 8 == 0)
      ? (int)(blockIdx.x % 8) * (total / 8) + (int)blockIdx.x / 8
      : (int)blockIdx.x;
  const int by = lin % GYB;
  const int bx = lin / GYB;
  const long b0 = (long)by * HF_BATCH;
  const long lab0 = (long)bx * HF_LABS + wave * 32;
  const int kj = (lane >> 4) * 8;
  const int NK = NKT ? NKT : EP / 32;

  // A rows = labels (W rows), B cols = batch rows (cv rows); both k-runs
  // contiguous.  AIMG: per-(row-tile, k-chunk) image reads instead.
  const long arow0 = lab0 + (lane & 15);
  const long arow1 = arow0 + 16;
  const bf16* ap0 = AIMG ? w + ((lab0 >> 4) * 4) * 512 + lane * 8
                         : w + arow0 * EP + kj;
  const bf16* ap1 = AIMG ? ap0 + 4 * 512 : w + arow1 * EP + kj;
  const bool a0ok = (AIMG || arow0 < L) && !(variant & 8);
  const bool a1ok = (AIMG || arow1 < L) && !(variant & 8);
  const long bcol = b0 + (lane & 15);  // + nt*16 per col-tile

  f32x4 acc[2][4];
#pragma unroll
  for (int rt = 0; rt < 2; ++rt)
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) acc[rt][nt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const bf16x8 zero8 = {};

  // LDS: one [64][264] bf16 plane, used as the block's cv tile during the
  // k-loop and REUSED as the output transpose patch in the epilogue
  // (barriers separate the lives).  264-elem (132-dword) row stride:
  // b128 fragment reads hit 16 distinct banks (4c mod 64, c = 0..15).
  __shared__ bf16 smem2d[HF_BATCH][HF_LABS + 8];
  __shared__ float wstat[8][HF_BATCH][2];

#define HF_LD(v, ok, p, kk)                                          \
  const bf16x8 v =                                                   \
      (ok) ? *(const bf16x8*)((p) + (AIMG ? (kk) * 512 : (kk) * 32)) \
           : zero8;
#define HF_LB(v, nt, kk) \
  const bf16x8 v = \
      *(const bf16x8*)&smem2d[nt * 16 + (lane & 15)][(kk)*32 + kj];
#define HF_MM(rt, nt, kk)                                       \
  acc[rt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(        \
      a##rt##kk, b##nt, acc[rt][nt], 0, 0, 0);

  if (NKT == 4) {
    // EP=128 fast path.  The cv tile (64 contiguous rows, 16 KB) is
    // staged block-cooperatively with full-line coalesced loads: per-wave
    // 16-B fragment loads from global scatter across 16 lines per
    // instruction and all 8 waves would issue them for the SAME tile
    // (measured 24 us of the kernel on their own).  W fragment loads
    // stay global — the tile is unique per block and already streams at
    // the HBM floor.  All 8 W loads issue before the MFMA chain (named
    // registers keep hipcc from demoting a staging array to scratch; a
    // load-in-loop schedule serializes on latency, measured 2x slower).
    if (!(variant & 4)) {
      for (int t = threadIdx.x; t < HF_BATCH * 16; t += 512) {
        const int r = t >> 4;
        const int c = t & 15;
        const bf16x8 v = (b0 + r < B)
            ? *(const bf16x8*)(cv + (b0 + r) * EP + c * 8) : zero8;
        *(bf16x8*)&smem2d[r][c * 8] = v;
      }
    }
    __syncthreads();
    HF_LD(a00, a0ok, ap0, 0) HF_LD(a01, a0ok, ap0, 1)
    HF_LD(a02, a0ok, ap0, 2) HF_LD(a03, a0ok, ap0, 3)
    HF_LD(a10, a1ok, ap1, 0) HF_LD(a11, a1ok, ap1, 1)
    HF_LD(a12, a1ok, ap1, 2) HF_LD(a13, a1ok, ap1, 3)
    {
      HF_LB(b0, 0, 0) HF_LB(b1, 1, 0) HF_LB(b2, 2, 0) HF_LB(b3, 3, 0)
      HF_MM(0, 0, 0) HF_MM(1, 0, 0) HF_MM(0, 1, 0) HF_MM(1, 1, 0)
      HF_MM(0, 2, 0) HF_MM(1, 2, 0) HF_MM(0, 3, 0) HF_MM(1, 3, 0)
    }
    {
      HF_LB(b0, 0, 1) HF_LB(b1, 1, 1) HF_LB(b2, 2, 1) HF_LB(b3, 3, 1)
      HF_MM(0, 0, 1) HF_MM(1, 0, 1) HF_MM(0, 1, 1) HF_MM(1, 1, 1)
      HF_MM(0, 2, 1) HF_MM(1, 2, 1) HF_MM(0, 3, 1) HF_MM(1, 3, 1)
    }
    {
      HF_LB(b0, 0, 2) HF_LB(b1, 1, 2) HF_LB(b2, 2, 2) HF_LB(b3, 3, 2)
      HF_MM(0, 0, 2) HF_MM(1, 0, 2) HF_MM(0, 1, 2) HF_MM(1, 1, 2)
      HF_MM(0, 2, 2) HF_MM(1, 2, 2) HF_MM(0, 3, 2) HF_MM(1, 3, 2)
    }
    {
      HF_LB(b0, 0, 3) HF_LB(b1, 1, 3) HF_LB(b2, 2, 3) HF_LB(b3, 3, 3)
      HF_MM(0, 0, 3) HF_MM(1, 0, 3) HF_MM(0, 1, 3) HF_MM(1, 1, 3)
      HF_MM(0, 2, 3) HF_MM(1, 2, 3) HF_MM(0, 3, 3) HF_MM(1, 3, 3)
    }
    // cv tile is dead from here; the epilogue below reuses smem2d as the
    // output patch after this barrier
    __syncthreads();
  } else {
    const bool b4ok = !(variant & 4);
#pragma unroll 4
    for (int kk = 0; kk < NK; ++kk) {
      const bf16x8 a0 =
          a0ok ? *(const bf16x8*)(ap0 + (long)kk * 32) : zero8;
      const bf16x8 a1 =
          a1ok ? *(const bf16x8*)(ap1 + (long)kk * 32) : zero8;
#pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const long bc = bcol + nt * 16;
        const bf16x8 b = (bc < B && b4ok)
            ? *(const bf16x8*)(cv + bc * EP + (long)kk * 32 + kj)
            : zero8;
        acc[0][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a0, b, acc[0][nt], 0, 0, 0);
        acc[1][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a1, b, acc[1][nt], 0, 0, 0);
      }
    }
  }
#undef HF_LD
#undef HF_LB
#undef HF_MM

  // epilogue: + bias, bf16 round, per-batch-row stats, and a block-wide
  // LDS transpose so global stores are full 128-B lines: each lane's 4
  // consecutive labels pack into one b64 LDS write (vs 32 bank-conflicted
  // b16 scalar writes for a row-major-orientation kernel — measured 2.6x
  // slower overall; see PERF.md), then 8 lanes stream each batch row out
  // in 16-B chunks.
  // 264-elem (528-B) row stride: b64 writes and b128 reads land 2-way
  // bank-conflicted (16-B-aligned strides can't do better) — negligible
  // at 8 writes + 16 reads per lane.
  bf16 (*patch)[HF_LABS + 8] = smem2d;  // k-loop cv tile, now dead
  float sm[4], ss[4];
#pragma unroll
  for (int nt = 0; nt < 4; ++nt) {
    sm[nt] = NINF;
    ss[nt] = 0.f;
  }

#pragma unroll
  for (int rt = 0; rt < 2; ++rt) {
    // this lane's 4 consecutive labels for this row-tile
    const long labb = lab0 + rt * 16 + (lane >> 4) * 4;
    const int labl = wave * 32 + rt * 16 + (lane >> 4) * 4;  // block-local
    float b4[4];
#pragma unroll
    for (int r = 0; r < 4; ++r)
      b4[r] = (labb + r < L) ? bias[labb + r] : 0.f;
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      bf16x4 v4;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const bf16 v = f2bf(acc[rt][nt][r] + b4[r]);
        v4[r] = v;
        if (STATS && labb + r < L) {
          // stats from the bf16-rounded value: exact parity with a loss
          // kernel that would re-read the stored logits
          const float xr = bf2f(v);
          if (xr > sm[nt]) {
            ss[nt] *= __expf(sm[nt] - xr);
            sm[nt] = xr;
          }
          ss[nt] += __expf(xr - sm[nt]);
        }
      }
      if (!(variant & 16))
        *(bf16x4*)&patch[nt * 16 + (lane & 15)][labl] = v4;
    }
  }

  if (STATS) {
    // combine across the 4 lanes (l, l^16, l^32, l^48) sharing a batch
    // row; lanes 0-15 stage this wave's 32-label partial for the
    // cross-wave merge below
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      float m = sm[nt], s = ss[nt];
#pragma unroll
      for (int off = 16; off < 64; off <<= 1) {
        const float om = __shfl_xor(m, off);
        const float os = __shfl_xor(s, off);
        HF_MERGE(m, s, om, os);
      }
      if (lane < 16) {
        wstat[wave][nt * 16 + lane][0] = m;
        wstat[wave][nt * 16 + lane][1] = s;
      }
    }
  }
  __syncthreads();

  // store: 8 batch rows per wave, 8 lanes x 16 B = one full 128-B line
  // per row per pass.  Nontemporal by default (logits far exceed L2);
  // C2V_HF_NT=0 uses normal stores so the tail of the matrix stays in L2
  // for the fused backward that reads logits right after (A/B knob —
  // the NT choice predates the recompute-G backward).
  const long cend = (L - (long)bx * HF_LABS < HF_LABS)
                        ? L - (long)bx * HF_LABS : HF_LABS;
  const int bcr = wave * 8 + (lane >> 3);
  const long grow = b0 + bcr;
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int lo = p * 64 + (lane & 7) * 8;
    if (grow < B && lo < cend && !(variant & 1)) {
      bf16* dst = out + grow * L + (long)bx * HF_LABS + lo;
      if (lo + 8 <= cend) {
        if (variant & 32)
          *(bf16x8*)dst = *(const bf16x8*)&patch[bcr][lo];
        else
          __builtin_nontemporal_store(*(const bf16x8*)&patch[bcr][lo],
                                      (bf16x8*)dst);
      } else {
        for (int j = 0; j < cend - lo; ++j) dst[j] = patch[bcr][lo + j];
      }
    }
  }

  if (STATS && wave == 0) {
    // merge the 8 per-wave partials -> one (m, s) per batch row per
    // 256-label block
    float m = NINF, s = 0.f;
#pragma unroll
    for (int wv = 0; wv < 8; ++wv) {
      const float om = wstat[wv][lane][0];
      const float os = wstat[wv][lane][1];
      HF_MERGE(m, s, om, os);
    }
    const long bc = b0 + lane;
    if (bc < B) {
      pm[(long)bx * B + bc] = m;
      ps[(long)bx * B + bc] = s;
    }
  }
}

// Merge the GXL per-label-block (m, s) partials of each batch row ->
// lse[b], and accumulate acc[0] += w_y*(lse - logit_y), acc[1] += w_y
// (same contract as lsm_nll_fwd_kernel in logsoftmax_nll.hip).  16 waves
// per block, one row per wave, block-reduced to ONE atomic pair per
// block — per-row atomics on 2 shared addresses serialize (PERF.md
// pathology #1).  Partials are laid out [lab_block][B] (coalesced
// producer writes); they were just written, so the strided reads here
// hit L2.
// acc_part[block][2] per-block partials (summed by slab_sum_f32 in fixed
// order): an atomic-add accumulation is ordering-nondeterministic at the
// fp32 ulp level, which made otherwise-identical training runs diverge
// after 2 steps of bf16 rounding.
__global__ __launch_bounds__(1024) void lsm_finalize_kernel(
    const bf16* __restrict__ logits, const float* __restrict__ pm,
    const float* __restrict__ ps, const long* __restrict__ label,
    const float* __restrict__ weight, float* __restrict__ lse,
    float* __restrict__ acc_part, int B, long L, int GXL) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int b = blockIdx.x * 16 + wave;
  __shared__ float red[16][2];
  float nll = 0.f, wsum = 0.f;
  if (b < B) {
    float m = NINF, s = 0.f;
    for (int j = lane; j < GXL; j += WAVE) {
      const float om = pm[(long)j * B + b];
      const float os = ps[(long)j * B + b];
      HF_MERGE(m, s, om, os);
    }
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      const float om = __shfl_xor(m, off);
      const float os = __shfl_xor(s, off);
      HF_MERGE(m, s, om, os);
    }
    if (lane == 0) {
      const float l = m + __logf(s);
      lse[b] = l;
      const long y = label[b];
      const float wy = weight ? weight[y] : 1.0f;
      nll = wy * (l - bf2f(logits[(long)b * L + y]));
      wsum = wy;
    }
  }
  if (lane == 0) {
    red[wave][0] = nll;
    red[wave][1] = wsum;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float a0 = 0.f, a1 = 0.f;
#pragma unroll
    for (int w = 0; w < 16; ++w) {
      a0 += red[w][0];
      a1 += red[w][1];
    }
    acc_part[(long)blockIdx.x * 2] = a0;
    acc_part[(long)blockIdx.x * 2 + 1] = a1;
  }
}

// W [L, 128] row-major -> MFMA A-fragment image [nrt][4 kc][64 lanes][8]:
// entry [rt][kc][l][j] = w[rt*16 + (l&15)][kc*32 + (l>>4)*8 + j]
// (zero-padded past L).  Makes head_fwd's A-operand reads contiguous
// 1-KB wave reads at large L, where the per-lane 256-B-strided row loads
// stream W from HBM inefficiently (variant-isolated ~80 us at L=261k).
__global__ __launch_bounds__(256) void swizzle_a_kernel(
    const bf16* __restrict__ w, bf16* __restrict__ wimg, long L) {
  const long rt = blockIdx.x;
  __shared__ bf16 tile[16][132];
  const bf16x8 zero8 = {};
  for (int t = threadIdx.x; t < 16 * 16; t += 256) {
    const int r = t >> 4;
    const int c = t & 15;
    const long row = rt * 16 + r;
    const bf16x8 v =
        (row < L) ? *(const bf16x8*)(w + row * 128 + c * 8) : zero8;
    *(bf16x8*)&tile[r][c * 8] = v;
  }
  __syncthreads();
  // 256 outputs of 16 B: (kc, lane) pairs, one per thread
  {
    const int kc = threadIdx.x >> 6;
    const int l = threadIdx.x & 63;
    bf16x8 v;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      v[j] = tile[l & 15][kc * 32 + (l >> 4) * 8 + j];
    *(bf16x8*)(wimg + ((rt * 4 + kc) * 64 + (long)l) * 8) = v;
  }
}

extern "C" {

void launch_swizzle_a(const void* w, void* wimg, long L, long nrt,
                      hipStream_t stream) {
  swizzle_a_kernel<<<(int)nrt, 256, 0, stream>>>((const bf16*)w,
                                                 (bf16*)wimg, L);
}

void launch_head_fwd(const void* cv, const void* w, const float* bias,
                     void* out, float* pm, float* ps, long B, long L, int EP,
                     int aimg, hipStream_t stream) {
  const int GXL = (int)((L + HF_LABS - 1) / HF_LABS);
  const int GYB = (int)((B + HF_BATCH - 1) / HF_BATCH);
  const dim3 grid((long)GXL * GYB);
  const int stats = pm != nullptr;
  // store mode by size: NORMAL stores up to ~the L2 scale (the fused
  // recompute-G backward reads logits right after; A/B 1.403 vs 1.407
  // ms/step at L=72k), NONTEMPORAL above it (write-allocate doubles the
  // 534-MB logits write traffic at L=261k).  C2V_HF_NT=0/1 overrides.
  static const int base_variant =
      getenv("C2V_HF_VARIANT") ? atoi(getenv("C2V_HF_VARIANT")) : 0;
  static const char* nt_env = getenv("C2V_HF_NT");
  const bool nt = nt_env ? (nt_env[0] == '1') : (L > 98304);
  const int variant = base_variant | (nt ? 0 : 32);
  // C2V_HF_OCC=1 reverts to the unconstrained register allocation
  static const char* occ_env = getenv("C2V_HF_OCC");
  const bool hf_occ1 = occ_env && occ_env[0] == '1';
#define HFCASE(nkt, st, ai)                                                  \
  if (hf_occ1)                                                               \
    head_fwd_kernel<nkt, st, ai, 1><<<grid, 512, 0, stream>>>(                \
        (const bf16*)cv, (const bf16*)w, bias, (bf16*)out, pm, ps, B, L,     \
        EP, GYB, variant);                                                    \
  else                                                                        \
    head_fwd_kernel<nkt, st, ai><<<grid, 512, 0, stream>>>(                   \
      (const bf16*)cv, (const bf16*)w, bias, (bf16*)out, pm, ps, B, L, EP,   \
      GYB, variant)
  if (EP == 128) {
    if (aimg) {
      if (stats) HFCASE(4, 1, 1); else HFCASE(4, 0, 1);
    } else {
      if (stats) HFCASE(4, 1, 0); else HFCASE(4, 0, 0);
    }
  } else {
    if (stats) HFCASE(0, 1, 0); else HFCASE(0, 0, 0);
  }
#undef HFCASE
}

void launch_lsm_finalize(const void* logits, const float* pm, const float* ps,
                         const long* label, const float* weight, float* lse,
                         float* acc, int B, long L, int GXL,
                         hipStream_t stream) {
  lsm_finalize_kernel<<<(B + 15) / 16, 1024, 0, stream>>>(
      (const bf16*)logits, pm, ps, label, weight, lse, acc, B, L, GXL);
}

}  // extern "C"
