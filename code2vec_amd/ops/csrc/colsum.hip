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

extern "C" {

void launch_colsum_bf16(const void* x, float* out, long B, long L,
                        hipStream_t stream) {
  const int rows_per_block = 32;  // wider grid: latency-bound otherwise
  const int lblocks = (int)((L + 255) / 256);
  const int rblocks = (int)((B + rows_per_block - 1) / rows_per_block);
  colsum_bf16_kernel<<<lblocks * rblocks, 256, 0, stream>>>(
      (const bf16*)x, out, B, L, rows_per_block);
}

}  // extern "C"
