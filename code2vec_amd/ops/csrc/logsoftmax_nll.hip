// K12: fused full-vocab log-softmax + weighted NLL, forward and backward.
// Reference math: main.py:251-264 with criterion nn.NLLLoss(weight=1/freq)
// (main.py:129-130): loss = sum_i w[y_i]*(lse_i - logit_i[y_i]) / sum_i w[y_i].
//
// Single pass over L per row (online max+sum), fp32 accumulation over bf16
// logits; built for L up to ~261k columns (java-large config).

#include "common.h"

// one block per row; online (max, sum) over the row's L columns
// acc_part[b][2] per-row partials (summed deterministically on the host
// side via slab_sum_f32 — see the ordering note in head_fwd.hip)
__global__ __launch_bounds__(256) void lsm_nll_fwd_kernel(
    const bf16* __restrict__ logits, const long* __restrict__ label,
    const float* __restrict__ weight, float* __restrict__ lse,
    float* __restrict__ acc_part, int B, long L) {
  const int b = blockIdx.x;
  const bf16* row = logits + (long)b * L;
  float m = -3.0e38f, s = 0.f;
  const long L8 = L & ~7L;
  for (long j = (long)threadIdx.x * 8; j < L8; j += (long)blockDim.x * 8) {
    bf16 v[8];
    *(uint4*)v = *(const uint4*)(row + j);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      const float x = bf2f(v[k]);
      if (x > m) {
        s *= __expf(m - x);
        m = x;
      }
      s += __expf(x - m);
    }
  }
  for (long j = L8 + threadIdx.x; j < L; j += blockDim.x) {
    const float x = bf2f(row[j]);
    if (x > m) {
      s *= __expf(m - x);
      m = x;
    }
    s += __expf(x - m);
  }
  // block-combine (m, s) pairs
  __shared__ float red_m[64], red_s[64];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float om = __shfl_xor(m, off);
    const float os = __shfl_xor(s, off);
    const float nm = fmaxf(m, om);
    s = s * __expf(m - nm) + os * __expf(om - nm);
    m = nm;
  }
  if (lane == 0) { red_m[wave] = m; red_s[wave] = s; }
  __syncthreads();
  if (threadIdx.x == 0) {
    const int nwaves = blockDim.x / WAVE;
    float M0 = red_m[0], S0 = red_s[0];
    for (int w = 1; w < nwaves; ++w) {
      const float nm = fmaxf(M0, red_m[w]);
      S0 = S0 * __expf(M0 - nm) + red_s[w] * __expf(red_m[w] - nm);
      M0 = nm;
    }
    const float l = M0 + __logf(S0);
    lse[b] = l;
    const long y = label[b];
    const float wy = weight ? weight[y] : 1.0f;
    acc_part[(long)b * 2] = wy * (l - bf2f(row[y]));
    acc_part[(long)b * 2 + 1] = wy;
  }
}

// 2-D-grid online-softmax partials: pm/ps[gx][b] = (max, sumexp) of
// logits[b, gx*CHUNK : (gx+1)*CHUNK).  The one-block-per-row kernel above
// walks L serially per block (B blocks only) — at L = 261k that is
// latency-bound at ~3.6 TB/s; this produces the same (pm, ps) layout the
// head-forward stats epilogue emits, merged by lsm_finalize_kernel
// (head_fwd.hip), and runs at the streaming floor.
#define LSMP_CHUNK 16384  // 32 KB/block: enough in-flight loads to hide latency
__global__ __launch_bounds__(256) void lsm_partial_kernel(
    const bf16* __restrict__ logits, float* __restrict__ pm,
    float* __restrict__ ps, int B, long L) {
  const int gxn = (int)((L + LSMP_CHUNK - 1) / LSMP_CHUNK);
  const int b = blockIdx.x / gxn;
  const int gx = blockIdx.x % gxn;
  const bf16* row = logits + (long)b * L + (long)gx * LSMP_CHUNK;
  const long n = min((long)LSMP_CHUNK, L - (long)gx * LSMP_CHUNK);
  // 8 independent (max, sum) accumulators, one per vector slot: the
  // single-accumulator form chains every element through a dependent
  // exp+compare (measured 4.9 TB/s); independent slots break the chain
  const long n8 = n & ~7L;
  float m8[8], s8[8];
#pragma unroll
  for (int k = 0; k < 8; ++k) { m8[k] = -3.0e38f; s8[k] = 0.f; }
  for (long j = (long)threadIdx.x * 8; j < n8; j += 256 * 8) {
    bf16 v[8];
    *(uint4*)v = *(const uint4*)(row + j);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      const float x = bf2f(v[k]);
      if (x > m8[k]) {
        s8[k] *= __expf(m8[k] - x);
        m8[k] = x;
      }
      s8[k] += __expf(x - m8[k]);
    }
  }
  float m = m8[0], s = s8[0];
#pragma unroll
  for (int k = 1; k < 8; ++k) {
    const float nm = fmaxf(m, m8[k]);
    s = s * __expf(m - nm) + s8[k] * __expf(m8[k] - nm);
    m = nm;
  }
  for (long j = n8 + threadIdx.x; j < n; j += 256) {
    const float x = bf2f(row[j]);
    if (x > m) {
      s *= __expf(m - x);
      m = x;
    }
    s += __expf(x - m);
  }
  __shared__ float red_m[4], red_s[4];
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float om = __shfl_xor(m, off);
    const float os = __shfl_xor(s, off);
    const float nm = fmaxf(m, om);
    s = s * __expf(m - nm) + os * __expf(om - nm);
    m = nm;
  }
  if (lane == 0) { red_m[wave] = m; red_s[wave] = s; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float M0 = red_m[0], S0 = red_s[0];
    for (int w = 1; w < 4; ++w) {
      const float nm = fmaxf(M0, red_m[w]);
      S0 = S0 * __expf(M0 - nm) + red_s[w] * __expf(red_m[w] - nm);
      M0 = nm;
    }
    pm[(long)gx * B + b] = M0;
    ps[(long)gx * B + b] = S0;
  }
}

// dlogits[b,j] = gscale * (w_y/wsum) * (exp(x - lse_b) - [j == y])
__global__ __launch_bounds__(256) void lsm_nll_bwd_kernel(
    const bf16* __restrict__ logits, const long* __restrict__ label,
    const float* __restrict__ weight, const float* __restrict__ lse,
    const float* __restrict__ acc, const float* __restrict__ gscale,
    bf16* __restrict__ dlogits, int B, long L) {
  const int b = blockIdx.x;
  const bf16* row = logits + (long)b * L;
  bf16* drow = dlogits + (long)b * L;
  const long y = label[b];
  const float wy = weight ? weight[y] : 1.0f;
  const float coef = gscale[0] * wy / acc[1];
  const float l = lse[b];
  const long L8 = L & ~7L;
  for (long j = (long)threadIdx.x * 8; j < L8; j += (long)blockDim.x * 8) {
    bf16 v[8];
    *(uint4*)v = *(const uint4*)(row + j);
    bf16 d[8];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float g = coef * __expf(bf2f(v[k]) - l);
      if (j + k == y) g -= coef;
      d[k] = f2bf(g);
    }
    *(uint4*)(drow + j) = *(uint4*)d;
  }
  for (long j = L8 + threadIdx.x; j < L; j += blockDim.x) {
    float g = coef * __expf(bf2f(row[j]) - l);
    if (j == y) g -= coef;
    drow[j] = f2bf(g);
  }
}

extern "C" {

void launch_lsm_nll_fwd(const void* logits, const long* label,
                        const float* weight, float* lse, float* acc, int B,
                        long L, hipStream_t stream) {
  lsm_nll_fwd_kernel<<<B, 256, 0, stream>>>((const bf16*)logits, label,
                                            weight, lse, acc, B, L);
}

void launch_lsm_partial(const void* logits, float* pm, float* ps, int B,
                        long L, hipStream_t stream) {
  const int gxn = (int)((L + LSMP_CHUNK - 1) / LSMP_CHUNK);
  lsm_partial_kernel<<<(long)B * gxn, 256, 0, stream>>>(
      (const bf16*)logits, pm, ps, B, L);
}

void launch_lsm_nll_bwd(const void* logits, const long* label,
                        const float* weight, const float* lse,
                        const float* acc, const float* gscale, void* dlogits,
                        int B, long L, hipStream_t stream) {
  lsm_nll_bwd_kernel<<<B, 256, 0, stream>>>((const bf16*)logits, label,
                                            weight, lse, acc, gscale,
                                            (bf16*)dlogits, B, L);
}

}  // extern "C"
