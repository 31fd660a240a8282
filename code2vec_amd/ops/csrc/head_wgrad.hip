// Output-head weight gradient: dW[L, EP] = dlogits^T[L, B] @ cv[B, EP]
// (B = batch = the K dim, 1024).  Both MFMA operands are k(=batch)-strided
// in memory, so each K-step stages 32-deep fragment images in LDS using the
// bank-spread 80-B column stride (see combiner.hip).  One block per 128-row
// L tile, 8 waves x 16 rows; dW writes are tiny (7.7 MB) so the epilogue
// stores scalar.

#include "common.h"

template <int NT>  // EP/16 column tiles (acc per wave: [NT][4])
__global__ __launch_bounds__(512) void head_wgrad_kernel(
    const bf16* __restrict__ dlogits, const bf16* __restrict__ cv,
    bf16* __restrict__ dw, long L, long B, int EP) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const long l0 = (long)blockIdx.x * 128;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // double-buffered fragment images, 80-B column stride (bank-spread for
  // both the scattered 2-B repack writes and the 16-B fragment reads)
  constexpr int ESTRIDE = 40;
  const size_t absz = (size_t)128 * ESTRIDE;
  const size_t bbsz = (size_t)EP * ESTRIDE;
  bf16* a_img = (bf16*)smem;              // [2][128][40]
  bf16* b_img = a_img + 2 * absz;         // [2][EP][40]

  f32x4 acc[NT];
#pragma unroll
  for (int n = 0; n < NT; ++n) acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kj = (lane >> 4) * 8;
  const int NK = (int)(B / 32);

  // per-thread staging chunks (named regs — arrays would spill):
  //   A: thread c -> dlogits[b = c/16][l0 + (c%16)*8 .. +8]
  //   Bv: thread c -> cv[b = c/(EP/8)][(c%(EP/8))*8 .. +8]
  const int a_brow = (int)threadIdx.x / 16;
  const int a_l8 = ((int)threadIdx.x % 16) * 8;
  const int b_chunks = 32 * EP / 8;
  const int b_brow = (int)threadIdx.x / (EP / 8);
  const int b_e8 = ((int)threadIdx.x % (EP / 8)) * 8;
  const bool b_active = (int)threadIdx.x < b_chunks;

  uint4 ra, rb;
  auto stage_load = [&](int kk) {
    uint4 va = {0, 0, 0, 0};
    if (l0 + a_l8 + 8 <= L)
      va = *(const uint4*)(dlogits + ((long)kk * 32 + a_brow) * L + l0 + a_l8);
    ra = va;
    if (b_active)
      rb = *(const uint4*)(cv + ((long)kk * 32 + b_brow) * EP + b_e8);
  };
  auto stage_write = [&](int buf) {
    bf16 v[8];
    *(uint4*)v = ra;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      a_img[buf * absz + (size_t)(a_l8 + j) * ESTRIDE + a_brow] = v[j];
    if (b_active) {
      *(uint4*)v = rb;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        b_img[buf * bbsz + (size_t)(b_e8 + j) * ESTRIDE + b_brow] = v[j];
    }
  };

  stage_load(0);
  stage_write(0);
  __syncthreads();
  for (int kk = 0; kk < NK; ++kk) {
    const int buf = kk & 1;
    if (kk + 1 < NK) stage_load(kk + 1);
    const bf16x8 a = *(const bf16x8*)(
        a_img + buf * absz +
        (size_t)(wave * 16 + (lane & 15)) * ESTRIDE + kj);
#pragma unroll
    for (int n = 0; n < NT; ++n) {
      const bf16x8 b = *(const bf16x8*)(
          b_img + buf * bbsz + (size_t)(n * 16 + (lane & 15)) * ESTRIDE + kj);
      acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[n], 0, 0, 0);
    }
    if (kk + 1 < NK) stage_write(buf ^ 1);
    __syncthreads();
  }

  // epilogue: C row = (lane>>4)*4 + r (+ wave*16), col = n*16 + (lane&15)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const long row = l0 + wave * 16 + (lane >> 4) * 4 + r;
    if (row < L) {
#pragma unroll
      for (int n = 0; n < NT; ++n) {
        dw[row * EP + n * 16 + (lane & 15)] = f2bf(acc[n][r]);
      }
    }
  }
}

extern "C" {

void launch_head_wgrad(const void* dlogits, const void* cv, void* dw, long L,
                       long B, int EP, hipStream_t stream) {
  const long grid = (L + 127) / 128;
  const int smem = 2 * (128 + EP) * 40 * (int)sizeof(bf16);
  const int NT = EP / 16;
#define HCASE(nt)                                                            \
  case nt:                                                                   \
    head_wgrad_kernel<nt><<<grid, 512, smem, stream>>>(                      \
        (const bf16*)dlogits, (const bf16*)cv, (bf16*)dw, L, B, EP);         \
    break;
  switch (NT) {
    HCASE(2) HCASE(4) HCASE(6) HCASE(8)
    default:
      printf("head_wgrad: unsupported EP=%d\n", EP);
  }
#undef HCASE
}

}  // extern "C"
