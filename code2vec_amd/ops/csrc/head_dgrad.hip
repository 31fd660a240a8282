// Output-head dgrad: dcv[B, EP] = dlogits[B, L] @ W[L, EP], split-K over
// the label dimension (K = L = 30k+) with fp32 partial slabs reduced by
// a single torch.sum on the host side (same recipe as the combiner
// wgrad, wgrad.hip, which beats hipBLASLt 3x on this skinny-output
// huge-K shape class; hipBLASLt runs this GEMM at ~89 us vs a ~15 us
// traffic floor).
//
// Takes W pre-transposed (Wt[EP, L], built once per step): with labels
// as the FAST axis of both operands every MFMA fragment is a contiguous
// 16-B run — dlogits fragments load straight from global, Wt sub-tiles
// stage into LDS with full-line coalesced loads and conflict-free b128
// fragment reads (132-elem padded row stride).  Without the transpose
// the W fragments would need a byte-scattered LDS transpose image (the
// expensive part of wgrad.hip's staging).
//
// Grid = (B/128) x ceil(L/512); block = 512 threads, 8 waves; wave owns
// 16 batch rows x all 8 EP col-tiles; each block accumulates its
// 512-label chunk in registers (4 staged sub-tiles of 128 labels) and
// writes one fp32 partial slab.  EP = 128 only (the generic-EP path
// falls back to rocBLAS in functional.py).

#include "common.h"

#define HD_CH 512    // labels per block (split-K chunk)
#define HD_SUB 128   // labels per staged LDS sub-tile
#define HD_BT 128    // batch rows per block

__global__ __launch_bounds__(512) void head_dgrad_kernel(
    const bf16* __restrict__ dlogits, const bf16* __restrict__ wt,
    float* __restrict__ partials, long B, long L, int GYB) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  // XCD-aware swizzle (see head_fwd.hip): consecutive blocks share a Wt
  // chunk; keep them on one XCD so the chunk is pulled from HBM once
  const int total = gridDim.x;
  const int lin = (total % 8 == 0)
      ? (int)(blockIdx.x % 8) * (total / 8) + (int)blockIdx.x / 8
      : (int)blockIdx.x;
  const int by = lin % GYB;
  const int sc = lin / GYB;
  const long bt0 = (long)by * HD_BT;
  const int kj = (lane >> 4) * 8;

  const long row = bt0 + wave * 16 + (lane & 15);
  const bool aok = row < B;
  const bf16* ap = dlogits + row * L + kj;

  // [128 EP rows][132] padded: b128 fragment reads hit 16 distinct banks
  __shared__ bf16 wst[128][HD_SUB + 4];

  f32x4 acc[8];
#pragma unroll
  for (int nt = 0; nt < 8; ++nt) acc[nt] = f32x4{0.f, 0.f, 0.f, 0.f};
  const bf16x8 zero8 = {};

  for (int s = 0; s < HD_CH / HD_SUB; ++s) {
    const long l0 = (long)sc * HD_CH + s * HD_SUB;
    // stage Wt[0..128][l0..l0+128) — coalesced 16-B chunks
    for (int t = threadIdx.x; t < 128 * (HD_SUB / 8); t += 512) {
      const int e = t >> 4;
      const int c = t & 15;
      const bf16x8 v = (l0 + c * 8 < L)
          ? *(const bf16x8*)(wt + (long)e * L + l0 + c * 8) : zero8;
      *(bf16x8*)&wst[e][c * 8] = v;
    }
    __syncthreads();
#pragma unroll
    for (int kk = 0; kk < HD_SUB / 32; ++kk) {
      const bf16x8 a = (aok && l0 + kk * 32 + kj < L)
          ? *(const bf16x8*)(ap + l0 + kk * 32) : zero8;
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

  // one fp32 partial slab per label chunk; reduced with torch.sum(dim=0)
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

// W transpose [L, 128] -> [128, L]: 64x64 LDS tiles, both global sides
// fully coalesced 16-B chunks (torch's transpose-copy runs at 0.8 TB/s,
// ~20 us for the 7.7 MB matrix — 4x off).
__global__ __launch_bounds__(256) void transpose_w_kernel(
    const bf16* __restrict__ w, bf16* __restrict__ wt, long L) {
  const int by = blockIdx.x & 1;         // e-tile (EP = 128 -> 2 tiles)
  const long bx = blockIdx.x >> 1;       // l-tile
  const long l0 = bx * 64;
  const int e0 = by * 64;
  __shared__ bf16 tile[64][72];
  for (int t = threadIdx.x; t < 512; t += 256) {
    const int r = t >> 3, c = t & 7;
    bf16x8 v = {};
    if (l0 + r < L)
      v = *(const bf16x8*)(w + (l0 + r) * 128 + e0 + c * 8);
    *(bf16x8*)&tile[r][c * 8] = v;
  }
  __syncthreads();
  for (int t = threadIdx.x; t < 512; t += 256) {
    const int r = t >> 3, c = t & 7;  // r = e within tile, c = l chunk
    bf16 v[8];
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = tile[c * 8 + j][r];
    if (l0 + c * 8 < L)  // L % 8 == 0 gate upstream
      *(bf16x8*)(wt + (long)(e0 + r) * L + l0 + c * 8) = *(bf16x8*)v;
  }
}

// partials[S][N] f32 -> out[N] bf16 (slab reduce + cast in one pass).
// Two independent accumulator chains per thread (i and i+stride): the
// serial per-thread load chain over S slabs was latency-bound at ~2 TB/s
// with 70+ slabs; doubling outstanding loads recovers most of it while
// keeping the full thread count (a wide-vector variant idles CUs).
__global__ __launch_bounds__(256) void slab_sum_bf16_kernel(
    const float* __restrict__ partials, bf16* __restrict__ out, int S,
    long N) {
  const long half = (N + 1) / 2;
  const long i = (long)blockIdx.x * 256 + threadIdx.x;
  if (i >= half) return;
  const long i2 = i + half;
  float acc = 0.f, acc2 = 0.f;
  for (int s = 0; s < S; ++s) {
    acc += __builtin_nontemporal_load(partials + s * N + i);
    if (i2 < N) acc2 += __builtin_nontemporal_load(partials + s * N + i2);
  }
  out[i] = f2bf(acc);
  if (i2 < N) out[i2] = f2bf(acc2);
}

extern "C" {

void launch_transpose_w(const void* w, void* wt, long L,
                        hipStream_t stream) {
  transpose_w_kernel<<<((L + 63) / 64) * 2, 256, 0, stream>>>(
      (const bf16*)w, (bf16*)wt, L);
}

void launch_slab_sum_bf16(const float* partials, void* out, int S, long N,
                          hipStream_t stream) {
  const long half = (N + 1) / 2;
  slab_sum_bf16_kernel<<<(half + 255) / 256, 256, 0, stream>>>(
      partials, (bf16*)out, S, N);
}

void launch_head_dgrad(const void* dlogits, const void* wt, float* partials,
                       long B, long L, hipStream_t stream) {
  const int GYB = (int)((B + HD_BT - 1) / HD_BT);
  const int SPLIT = (int)((L + HD_CH - 1) / HD_CH);
  head_dgrad_kernel<<<GYB * SPLIT, 512, 0, stream>>>(
      (const bf16*)dlogits, (const bf16*)wt, partials, B, L, GYB);
}

}  // extern "C"
