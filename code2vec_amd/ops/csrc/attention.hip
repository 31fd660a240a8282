// K7-K9 fused single-query attention + weighted pool, forward and backward.
// Reference math: model/model.py:62-69,90-105 —
//   scores[b,c] = sum_e ccv[b,c,e] * a[e]
//   scores = scores*mask + (1-mask)*NINF      (mask = starts>0; NINF=-3.4e38)
//   attn = softmax(scores, dim=C)
//   cv[b,e] = sum_c attn[b,c] * ccv[b,c,e]
//
// One workgroup per batch row; the [C, EP] ccv slice is staged in LDS once
// and reused by the score and pool phases (C=200, EP=128 -> 50 KiB).

#include "common.h"

#define NINF_F (-3.4e38f)

// ---------------------------------------------------------------------------
template <bool STAGE_LDS>
__global__ __launch_bounds__(256) void attention_fwd_kernel(
    const bf16* __restrict__ ccv, const float* __restrict__ a,
    const int* __restrict__ starts, float* __restrict__ cv,
    float* __restrict__ attn, int B, int C, int EP, int E) {
  const int b = blockIdx.x;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // layout: [C*EP bf16 tile (optional)] [EP f32 a] [C f32 scores] [64 f32 red]
  bf16* tile = (bf16*)smem;
  float* lds_a = (float*)(smem + (STAGE_LDS ? (size_t)C * EP * 2 : 0));
  float* scores = lds_a + EP;
  float* red = scores + C;

  const bf16* src = ccv + (long)b * C * EP;
  for (int e = threadIdx.x; e < EP; e += blockDim.x) lds_a[e] = a[e];
  if (STAGE_LDS) {
    const int total = C * EP / 8;
    const uint4* s4 = (const uint4*)src;
    uint4* d4 = (uint4*)tile;
    for (int i = threadIdx.x; i < total; i += blockDim.x) d4[i] = s4[i];
  }
  __syncthreads();

  // phase 1: scores (thread-per-context dot over EP; pad cols of a are 0)
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    const bf16* row = (STAGE_LDS ? tile : src) + (long)c * EP;
    float s = 0.f;
    for (int e = 0; e < EP; e += 8) {
      bf16 v[8];
      *(uint4*)v = *(const uint4*)(row + e);
#pragma unroll
      for (int j = 0; j < 8; ++j) s += bf2f(v[j]) * lds_a[e + j];
    }
    const float mask = starts[(long)b * C + c] > 0 ? 1.0f : 0.0f;
    scores[c] = s * mask + (1.0f - mask) * NINF_F;
  }
  __syncthreads();

  // phase 2: block softmax over C
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
    attn[(long)b * C + c] = scores[c];
  }
  __syncthreads();

  // phase 3: cv[e] = sum_c attn[c]*ccv[c,e] (thread-per-column, coalesced)
  for (int e = threadIdx.x; e < EP; e += blockDim.x) {
    float acc = 0.f;
    const bf16* base = (STAGE_LDS ? tile : src) + e;
    for (int c = 0; c < C; ++c) acc += scores[c] * bf2f(base[(long)c * EP]);
    cv[(long)b * EP + e] = acc;
  }
}

// ---------------------------------------------------------------------------
// Backward.  g[c] = dot(dcv, ccv[c]) (+ dattn[c]); sum_g = sum_c attn[c]*g[c];
// ds[c] = attn[c]*(g[c]-sum_g);  dccv[c,e] = attn[c]*dcv[e] + ds[c]*mask*a[e];
// da[e] += sum_c ds[c]*mask[c]*ccv[c,e].
template <bool STAGE_LDS>
__global__ __launch_bounds__(256) void attention_bwd_kernel(
    const float* __restrict__ dcv, const float* __restrict__ dattn,
    const bf16* __restrict__ ccv, const float* __restrict__ a,
    const int* __restrict__ starts, const float* __restrict__ attn,
    bf16* __restrict__ dccv, float* __restrict__ da, int B, int C, int EP,
    int E, int has_dattn) {
  const int b = blockIdx.x;
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16* tile = (bf16*)smem;
  float* lds_dcv = (float*)(smem + (STAGE_LDS ? (size_t)C * EP * 2 : 0));
  float* lds_a = lds_dcv + EP;
  float* ds = lds_a + EP;     // [C]
  float* red = ds + C;        // [64]

  const bf16* src = ccv + (long)b * C * EP;
  for (int e = threadIdx.x; e < EP; e += blockDim.x) {
    lds_dcv[e] = dcv[(long)b * EP + e];
    lds_a[e] = a[e];
  }
  if (STAGE_LDS) {
    const int total = C * EP / 8;
    const uint4* s4 = (const uint4*)src;
    uint4* d4 = (uint4*)tile;
    for (int i = threadIdx.x; i < total; i += blockDim.x) d4[i] = s4[i];
  }
  __syncthreads();

  // g[c] and partial sum_g
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int nwaves = blockDim.x / WAVE;
  float part = 0.f;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    const bf16* row = (STAGE_LDS ? tile : src) + (long)c * EP;
    float g = 0.f;
    for (int e = 0; e < EP; e += 8) {
      bf16 v[8];
      *(uint4*)v = *(const uint4*)(row + e);
#pragma unroll
      for (int j = 0; j < 8; ++j) g += bf2f(v[j]) * lds_dcv[e + j];
    }
    if (has_dattn) g += dattn[(long)b * C + c];
    const float at = attn[(long)b * C + c];
    ds[c] = g;  // temporarily g
    part += at * g;
  }
  part = wave_reduce_sum(part);
  if (lane == 0) red[wave] = part;
  __syncthreads();
  if (wave == 0) {
    float v = lane < nwaves ? red[lane] : 0.f;
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

  // dccv + da partials (thread-per-column)
  for (int e = threadIdx.x; e < EP; e += blockDim.x) {
    const float dcv_e = lds_dcv[e];
    const float a_e = lds_a[e];
    float da_acc = 0.f;
    const bf16* base = (STAGE_LDS ? tile : src) + e;
    bf16* dbase = dccv + (long)b * C * EP + e;
    for (int c = 0; c < C; ++c) {
      const float at = attn[(long)b * C + c];
      const float d = at * dcv_e + ds[c] * a_e;
      dbase[(long)c * EP] = f2bf(d);
      da_acc += ds[c] * bf2f(base[(long)c * EP]);
    }
    // per-block partials row (summed host-side; avoids same-address atomics)
    da[(long)b * EP + e] = e < E ? da_acc : 0.f;
  }
}

extern "C" {

static size_t attn_smem(int C, int EP, bool stage, bool bwd) {
  size_t s = stage ? (size_t)C * EP * 2 : 0;
  s += (size_t)EP * 4;            // a (fwd) / dcv (bwd)
  if (bwd) s += (size_t)EP * 4;   // a (bwd extra)
  s += (size_t)C * 4;             // scores / ds
  s += 64 * 4;                    // reduction scratch
  return s;
}

void launch_attention_fwd(const void* ccv, const float* a, const int* starts,
                          float* cv, float* attn, int B, int C, int EP, int E,
                          hipStream_t stream) {
  bool stage = attn_smem(C, EP, true, false) <= 160 * 1024 - 1024;
  size_t smem = attn_smem(C, EP, stage, false);
  if (stage)
    attention_fwd_kernel<true><<<B, 256, smem, stream>>>(
        (const bf16*)ccv, a, starts, cv, attn, B, C, EP, E);
  else
    attention_fwd_kernel<false><<<B, 256, smem, stream>>>(
        (const bf16*)ccv, a, starts, cv, attn, B, C, EP, E);
}

void launch_attention_bwd(const float* dcv, const float* dattn,
                          const void* ccv, const float* a, const int* starts,
                          const float* attn, void* dccv, float* da, int B,
                          int C, int EP, int E, int has_dattn,
                          hipStream_t stream) {
  bool stage = attn_smem(C, EP, true, true) <= 160 * 1024 - 1024;
  size_t smem = attn_smem(C, EP, stage, true);
  if (stage)
    attention_bwd_kernel<true><<<B, 256, smem, stream>>>(
        dcv, dattn, (const bf16*)ccv, a, starts, attn, (bf16*)dccv, da, B, C,
        EP, E, has_dattn);
  else
    attention_bwd_kernel<false><<<B, 256, smem, stream>>>(
        dcv, dattn, (const bf16*)ccv, a, starts, attn, (bf16*)dccv, da, B, C,
        EP, E, has_dattn);
}

}  // extern "C"
