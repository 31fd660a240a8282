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

// K17: row max+argmax over bf16 logits (eval-path prediction).  One block
// per row; ties resolve to the SMALLEST index (torch.max semantics).
__global__ __launch_bounds__(256) void row_max_argmax_kernel(
    const bf16* __restrict__ x, float* __restrict__ vals,
    long* __restrict__ idx, long L) {
  const bf16* row = x + (long)blockIdx.x * L;
  float best = -3.4e38f;
  long bi = 0;
  for (long j = threadIdx.x; j < L; j += 256) {
    const float v = bf2f(row[j]);
    if (v > best || (v == best && j < bi)) {
      best = v;
      bi = j;
    }
  }
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    const float ov = __shfl_xor(best, off);
    const long oi = (long)__shfl_xor((long long)bi, off);
    if (ov > best || (ov == best && oi < bi)) {
      best = ov;
      bi = oi;
    }
  }
  __shared__ float rv[4];
  __shared__ long ri[4];
  if (lane == 0) {
    rv[wave] = best;
    ri[wave] = bi;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    for (int w = 1; w < 4; ++w) {
      if (rv[w] > best || (rv[w] == best && ri[w] < bi)) {
        best = rv[w];
        bi = ri[w];
      }
    }
    vals[blockIdx.x] = best;
    idx[blockIdx.x] = bi;
  }
}

extern "C" {

void launch_row_max_argmax(const void* x, float* vals, long* idx, long B,
                           long L, hipStream_t stream) {
  row_max_argmax_kernel<<<B, 256, 0, stream>>>((const bf16*)x, vals, idx,
                                               L);
}

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
