// K7-K9 fused single-query attention + weighted pool, forward and backward.
// Reference math: model/model.py:62-69,90-105 —
//   scores[b,c] = sum_e ccv[b,c,e] * a[e]
//   scores = scores*mask + (1-mask)*NINF      (mask = starts>0; NINF=-3.4e38)
//   attn = softmax(scores, dim=C)
//   cv[b,e] = sum_c attn[b,c] * ccv[b,c,e]
//
// One workgroup (256 threads = 4 waves) per batch row; the [C, EP] ccv
// slice is staged in LDS once and reused by every phase.  Work is
// row-oriented: each 16-lane group owns one context at a time (8 columns
// per lane -> bf16x8 vector accesses, coalesced global writes in backward).

#include "common.h"

#define NINF_F (-3.4e38f)

// block softmax over scores[C] (in LDS), returns nothing (scores -> attn)
__device__ __forceinline__ void block_softmax(float* scores, float* red,
                                              float* attn_out, int C,
                                              long base) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int nwaves = blockDim.x / WAVE;
  float m = NINF_F;
  for (int c = threadIdx.x; c < C; c += blockDim.x) m = fmaxf(m, scores[c]);
  m = wave_reduce_max(m);
  if (lane == 0) red[wave] = m;
  __syncthreads();
  if (wave == 0) {
    float v = lane < nwaves ? red[lane] : NINF_F;
    v = wave_reduce_max(v);
    if (lane == 0) red[0] = v;
  }
  __syncthreads();
  m = red[0];
  float ssum = 0.f;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    const float e = __expf(scores[c] - m);
    scores[c] = e;
    ssum += e;
  }
  ssum = wave_reduce_sum(ssum);
  if (lane == 0) red[8 + wave] = ssum;
  __syncthreads();
  if (wave == 0) {
    float v = lane < nwaves ? red[8 + lane] : 0.f;
    v = wave_reduce_sum(v);
    if (lane == 0) red[8] = v;
  }
  __syncthreads();
  const float inv_sum = 1.0f / red[8];
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    scores[c] *= inv_sum;
    attn_out[base + c] = scores[c];
  }
  __syncthreads();
}

// per-16-lane-group dot of a ccv row against a vector held in LDS
// (each lane covers cols (l&15)*8 + j + 128*i)
__device__ __forceinline__ float group_row_dot(const bf16* row,
                                               const float* vec, int EP,
                                               int lane16) {
  float s = 0.f;
  for (int e0 = lane16 * 8; e0 < EP; e0 += 128) {
    bf16 v[8];
    *(uint4*)v = *(const uint4*)(row + e0);
#pragma unroll
    for (int j = 0; j < 8; ++j) s += bf2f(v[j]) * vec[e0 + j];
  }
  return s;
}

// ---------------------------------------------------------------------------
// ts = round8(E) <= EP: only the valid columns are staged/dotted — the
// EP-pad columns of ccv are zero and a/dcv pad is zero, so they never
// contribute; trimming them keeps the tile under the 53.3-KB line for
// 3 blocks/CU (54.8 KB = 2 blocks, a 33% occupancy loss).  Outputs in
// [ts, EP) are written as explicit zeros.
template <bool STAGE_LDS>
__global__ __launch_bounds__(256) void attention_fwd_kernel(
    const bf16* __restrict__ ccv, const float* __restrict__ a,
    const int* __restrict__ starts, float* __restrict__ cv,
    float* __restrict__ attn, int B, int C, int EP, int E, int ts) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // layout: [ccv tile (opt)] [a f32 EP] [scores C] [red 64] [partials 4*EP]
  bf16* tile = (bf16*)smem;
  float* lds_a = (float*)(smem + (STAGE_LDS ? (size_t)C * ts * 2 : 0));
  float* scores = lds_a + EP;
  float* red = scores + C;
  float* partials = red + 64;
  const int rstride = STAGE_LDS ? ts : EP;  // tile vs global row stride

  // persistent grid: only 3 blocks fit per CU at this LDS size, so a
  // block-per-method grid (B=1024) runs an underfilled tail round;
  // looping methods over occupancy*CU blocks balances across CUs
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
  const bf16* src = ccv + (long)b * C * EP;
  for (int e = threadIdx.x; e < EP; e += blockDim.x) lds_a[e] = a[e];
  if (STAGE_LDS) {
    const int row_chunks = ts / 8;
    const int total = C * row_chunks;
    const uint4* s4 = (const uint4*)src;
    uint4* d4 = (uint4*)tile;
    for (int i = threadIdx.x; i < total; i += blockDim.x)
      d4[i] = s4[(i / row_chunks) * (EP / 8) + i % row_chunks];
  }
  __syncthreads();

  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int lane16 = lane & 15;
  const int group = (threadIdx.x >> 4);  // 16 groups of 16 lanes

  // phase 1: scores — each 16-lane group owns one context per iteration
  for (int c = group; c < C; c += 16) {
    const bf16* row = (STAGE_LDS ? tile : src) + (long)c * rstride;
    float s = group_row_dot(row, lds_a, ts, lane16);
    s = group16_reduce_sum(s);
    if (lane16 == 0) {
      const float mask = starts[(long)b * C + c] > 0 ? 1.0f : 0.0f;
      scores[c] = s * mask + (1.0f - mask) * NINF_F;
    }
  }
  __syncthreads();

  // phase 2: softmax (scores -> attn probabilities, also written out)
  block_softmax(scores, red, attn, C, (long)b * C);

  // phase 3: cv[e] = sum_c attn[c]*ccv[c,e] — each wave accumulates a
  // contiguous context span into per-lane column pairs, then cross-wave sum
  const int span = (C + 3) / 4;
  const int c_lo = wave * span;
  const int c_hi = min(C, c_lo + span);
  for (int e0 = lane * 2; e0 < EP; e0 += WAVE * 2) {
    float acc0 = 0.f, acc1 = 0.f;
    if (e0 < ts) {
      for (int c = c_lo; c < c_hi; ++c) {
        const bf16* row = (STAGE_LDS ? tile : src) + (long)c * rstride;
        const bf16x2 v = *(const bf16x2*)(row + e0);
        const float at = scores[c];
        acc0 += at * bf2f(v[0]);
        acc1 += at * bf2f(v[1]);
      }
    }
    partials[wave * EP + e0] = acc0;
    partials[wave * EP + e0 + 1] = acc1;
  }
  __syncthreads();
  for (int e = threadIdx.x; e < EP; e += blockDim.x) {
    cv[(long)b * EP + e] = partials[e] + partials[EP + e] +
                           partials[2 * EP + e] + partials[3 * EP + e];
  }
  __syncthreads();  // smem reused by the next method
  }
}

// ---------------------------------------------------------------------------
// Backward.  g[c] = dot(dcv, ccv[c]) (+ dattn[c]); sum_g = sum_c attn[c]*g[c];
// ds[c] = attn[c]*(g[c]-sum_g)*mask;  dccv[c,e] = attn[c]*dcv[e] + ds[c]*a[e];
// da[e] = sum_c ds[c]*ccv[c,e]  (per-block partials row, summed host-side).
template <bool STAGE_LDS>
__global__ __launch_bounds__(256) void attention_bwd_kernel(
    const float* __restrict__ dcv, const float* __restrict__ dattn,
    const bf16* __restrict__ ccv, const float* __restrict__ a,
    const int* __restrict__ starts, const float* __restrict__ attn,
    bf16* __restrict__ dccv, float* __restrict__ da, int B, int C, int EP,
    int E, int has_dattn, int ts) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* tile = (bf16*)smem;
  float* lds_dcv = (float*)(smem + (STAGE_LDS ? (size_t)C * ts * 2 : 0));
  float* lds_a = lds_dcv + EP;
  float* ds = lds_a + EP;        // [C]
  float* red = ds + C;           // [64]
  float* partials = red + 64;    // [4][EP] (da)
  const int rstride = STAGE_LDS ? ts : EP;

  // persistent grid (see attention_fwd_kernel)
  for (int b = blockIdx.x; b < B; b += gridDim.x) {
  const bf16* src = ccv + (long)b * C * EP;
  for (int e = threadIdx.x; e < EP; e += blockDim.x) {
    lds_dcv[e] = dcv[(long)b * EP + e];
    lds_a[e] = a[e];
  }
  if (STAGE_LDS) {
    const int row_chunks = ts / 8;
    const int total = C * row_chunks;
    const uint4* s4 = (const uint4*)src;
    uint4* d4 = (uint4*)tile;
    for (int i = threadIdx.x; i < total; i += blockDim.x)
      d4[i] = s4[(i / row_chunks) * (EP / 8) + i % row_chunks];
  }
  __syncthreads();

  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int lane16 = lane & 15;
  const int group = (threadIdx.x >> 4);

  // phase A: g[c] (group-per-context) and block-reduced sum_g
  float part = 0.f;
  for (int c = group; c < C; c += 16) {
    const bf16* row = (STAGE_LDS ? tile : src) + (long)c * rstride;
    float g = group_row_dot(row, lds_dcv, ts, lane16);
    g = group16_reduce_sum(g);
    if (lane16 == 0) {
      if (has_dattn) g += dattn[(long)b * C + c];
      ds[c] = g;  // temporarily g
      part += attn[(long)b * C + c] * g;
    }
  }
  part = wave_reduce_sum(part);
  if (lane == 0) red[wave] = part;
  __syncthreads();
  if (wave == 0) {
    float v = lane < (blockDim.x / WAVE) ? red[lane] : 0.f;
    v = wave_reduce_sum(v);
    if (lane == 0) red[0] = v;
  }
  __syncthreads();
  const float sum_g = red[0];
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    const float at = attn[(long)b * C + c];
    const float mask = starts[(long)b * C + c] > 0 ? 1.0f : 0.0f;
    ds[c] = at * (ds[c] - sum_g) * mask;
  }
  __syncthreads();

  // phase B: dccv rows (group-per-context, coalesced bf16x8 writes) and
  // per-lane da partial accumulators
  // da accumulation: this lane's columns are (lane16*8 + j) + 128*i;
  // statically unrolled i (dynamic array indexing would go to scratch)
  float da_acc[3][8] = {};  // EP <= 384
  for (int c = group; c < C; c += 16) {
    const float at = attn[(long)b * C + c];
    const float dsc = ds[c];
    const bf16* row = (STAGE_LDS ? tile : src) + (long)c * rstride;
    bf16* drow = dccv + ((long)b * C + c) * EP;
#pragma unroll
    for (int i = 0; i < 3; ++i) {
      const int e0 = lane16 * 8 + 128 * i;
      if (e0 < ts) {
        bf16 v[8], o[8];
        *(uint4*)v = *(const uint4*)(row + e0);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          o[j] = f2bf(at * lds_dcv[e0 + j] + dsc * lds_a[e0 + j]);
          da_acc[i][j] += dsc * bf2f(v[j]);
        }
        *(uint4*)(drow + e0) = *(uint4*)o;
      } else if (e0 < EP) {
        // pad columns: dcv/a pad is zero, so the grad is exactly zero
        const uint4 z = {0, 0, 0, 0};
        *(uint4*)(drow + e0) = z;
      }
    }
  }
  // combine da partials DETERMINISTICALLY: the 4 lane-groups of each
  // wave that share lane16 combine via a fixed shuffle tree, each wave
  // writes its own partials row, and the final 4-way sum runs in fixed
  // wave order.  (The previous LDS atomicAdd combine was ordering-
  // nondeterministic at the fp32 ulp — identical runs diverged after a
  // couple of bf16-rounded optimizer steps.)
  __syncthreads();  // ds/tile reads done; reuse partials region
  for (int e = threadIdx.x; e < 4 * EP; e += blockDim.x)
    if ((e % EP) >= ts) partials[e] = 0.f;  // tail cols no lane writes
#pragma unroll
  for (int i = 0; i < 3; ++i) {
    const int e_base = lane16 * 8 + 128 * i;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = da_acc[i][j];
      v += __shfl_xor(v, 16);
      v += __shfl_xor(v, 32);
      if (lane < 16 && e_base < ts) partials[wave * EP + e_base + j] = v;
    }
  }
  __syncthreads();
  for (int e = threadIdx.x; e < EP; e += blockDim.x) {
    const float s4v = partials[e] + partials[EP + e] + partials[2 * EP + e] +
                      partials[3 * EP + e];
    da[(long)b * EP + e] = e < E ? s4v : 0.f;
  }
  __syncthreads();  // smem reused by the next method
  }
}

// grid = min(B, occupancy * CUs): just enough resident blocks that the
// persistent method loop balances across CUs with no tail round
template <typename K>
static int attn_grid(K kern, size_t smem, int B) {
  static int ncu = 0;
  if (ncu == 0)
    hipDeviceGetAttribute(&ncu, hipDeviceAttributeMultiprocessorCount, 0);
  int occ = 0;
  hipOccupancyMaxActiveBlocksPerMultiprocessor(&occ, kern, 256, smem);
  if (occ < 1) occ = 1;
  const long g = (long)occ * (ncu > 0 ? ncu : 256);
  return (int)(g < B ? g : B);
}

extern "C" {

static size_t attn_smem(int C, int EP, int ts, bool stage, bool bwd) {
  size_t s = stage ? (size_t)C * ts * 2 : 0;
  s += (size_t)EP * 4;            // a (fwd) / dcv (bwd)
  if (bwd) s += (size_t)EP * 4;   // a (bwd extra)
  s += (size_t)C * 4;             // scores / ds
  s += 64 * 4;                    // reduction scratch
  s += (size_t)4 * EP * 4;        // partials
  return s;
}

static int attn_ts(int E, int EP) {
  int ts = (E + 7) & ~7;
  return ts > EP ? EP : ts;
}

void launch_attention_fwd(const void* ccv, const float* a, const int* starts,
                          float* cv, float* attn, int B, int C, int EP, int E,
                          hipStream_t stream) {
  const int ts = attn_ts(E, EP);
  bool stage = attn_smem(C, EP, ts, true, false) <= 160 * 1024 - 1024;
  size_t smem = attn_smem(C, EP, ts, stage, false);
  if (stage)
    attention_fwd_kernel<true>
        <<<attn_grid(attention_fwd_kernel<true>, smem, B), 256, smem,
           stream>>>((const bf16*)ccv, a, starts, cv, attn, B, C, EP, E, ts);
  else
    attention_fwd_kernel<false>
        <<<attn_grid(attention_fwd_kernel<false>, smem, B), 256, smem,
           stream>>>((const bf16*)ccv, a, starts, cv, attn, B, C, EP, E, ts);
}

void launch_attention_bwd(const float* dcv, const float* dattn,
                          const void* ccv, const float* a, const int* starts,
                          const float* attn, void* dccv, float* da, int B,
                          int C, int EP, int E, int has_dattn,
                          hipStream_t stream) {
  const int ts = attn_ts(E, EP);
  bool stage = attn_smem(C, EP, ts, true, true) <= 160 * 1024 - 1024;
  size_t smem = attn_smem(C, EP, ts, stage, true);
  if (stage)
    attention_bwd_kernel<true>
        <<<attn_grid(attention_bwd_kernel<true>, smem, B), 256, smem,
           stream>>>(dcv, dattn, (const bf16*)ccv, a, starts, attn,
                     (bf16*)dccv, da, B, C, EP, E, has_dattn, ts);
  else
    attention_bwd_kernel<false>
        <<<attn_grid(attention_bwd_kernel<false>, smem, B), 256, smem,
           stream>>>(dcv, dattn, (const bf16*)ccv, a, starts, attn,
                     (bf16*)dccv, da, B, C, EP, E, has_dattn, ts);
}

}  // extern "C"
