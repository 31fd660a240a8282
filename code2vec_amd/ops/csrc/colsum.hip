// Column sum of a bf16 [B, L] matrix -> fp32 [L] (output-head bias grad:
// dbias = sum_b dlogits[b, :]).  torch's generic reduce runs this at
// ~1 TB/s; a simple row-looped, column-parallel kernel streams at HBM BW.
// Blocks tile (L-chunk x row-chunk); row-chunk partials combine with one
// fp32 atomic per column per row-chunk (few per column).

#include "common.h"

__global__ __launch_bounds__(256) void colsum_bf16_kernel(
    const bf16* __restrict__ x, float* __restrict__ out, long B, long L,
    int rows_per_block) {
  const long col = (long)(blockIdx.x % ((L + 255) / 256)) * 256 + threadIdx.x;
  const int rchunk = blockIdx.x / ((L + 255) / 256);
  if (col >= L) return;
  const long r0 = (long)rchunk * rows_per_block;
  const long r1 = min(r0 + rows_per_block, B);
  float acc = 0.f;
  for (long b = r0; b < r1; ++b) acc += bf2f(x[b * L + col]);
  atomic_add_f32(out + col, acc);
}

// Tiny partial-slab reduce: p[S, N] f32 -> out[N] f32, one block per
// column (N is small: EP/KP-scale; S <= ~1024 reduction-partial rows).
// torch's generic reduce spends 10 us per call on these; this is ~2 us.
__global__ __launch_bounds__(256) void slab_sum_f32_kernel(
    const float* __restrict__ p, float* __restrict__ out, int S, int N) {
  const int col = blockIdx.x;
  float acc = 0.f;
  for (int s = threadIdx.x; s < S; s += 256)
    acc += p[(long)s * N + col];
  acc = wave_reduce_sum(acc);
  __shared__ float red[4];
  const int wave = threadIdx.x / WAVE;
  if ((threadIdx.x & (WAVE - 1)) == 0) red[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) out[col] = red[0] + red[1] + red[2] + red[3];
}

extern "C" {

void launch_slab_sum_f32(const float* p, float* out, int S, int N,
                         hipStream_t stream) {
  slab_sum_f32_kernel<<<N, 256, 0, stream>>>(p, out, S, N);
}

void launch_colsum_bf16(const void* x, float* out, long B, long L,
                        hipStream_t stream) {
  const int rows_per_block = 32;  // wider grid: latency-bound otherwise
  const int lblocks = (int)((L + 255) / 256);
  const int rblocks = (int)((B + rows_per_block - 1) / rows_per_block);
  colsum_bf16_kernel<<<lblocks * rblocks, 256, 0, stream>>>(
      (const bf16*)x, out, B, L, rows_per_block);
}

}  // extern "C"
