// K14 (dgrad): dX[M, KP] = dZ[M, EP] @ W2[KP, EP]^T-ish — the combiner
// input gradient.  W2 is the [KP, EP] transpose of the stored [EP, KP]
// weight (98 KB, transposed host-side per step), so the MFMA B fragment
// B[k=ep][col=kp] = W2[kp][ep..ep+8] is memory-contiguous; the A fragment
// dZ[m][ep..ep+8] is contiguous too — no LDS staging for operands.
//
// Block = 512 threads = 8 waves; wave w owns rows [w*16, w*16+16) of a
// 128-row tile and the FULL KP width (acc[KP/16][4] fp32).  The epilogue
// bounces through a per-wave LDS half-tile for coalesced 16-B stores.

#include "common.h"

template <int NTK>  // KP/16 column tiles
__global__ __launch_bounds__(512) void dgrad_kernel(
    const bf16* __restrict__ dZ, const bf16* __restrict__ W2,
    bf16* __restrict__ dX, long M, int EP) {
  const int KP = NTK * 16;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const long row0 = (long)blockIdx.x * 128 + wave * 16;

  f32x4 acc[NTK];
#pragma unroll
  for (int n = 0; n < NTK; ++n) acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};

  long arow = row0 + (lane & 15);
  if (arow >= M) arow = M - 1;
  const int kj = (lane >> 4) * 8;
  const bf16* wrow = W2 + (size_t)(lane & 15) * EP + kj;

  const int NK = EP / 32;
  for (int kk = 0; kk < NK; ++kk) {
    const bf16x8 a = *(const bf16x8*)(dZ + arow * EP + kk * 32 + kj);
#pragma unroll
    for (int n = 0; n < NTK; ++n) {
      const bf16x8 b =
          *(const bf16x8*)(wrow + (size_t)n * 16 * EP + kk * 32);
      acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[n], 0, 0, 0);
    }
  }

  // epilogue: two half-width LDS bounces -> coalesced 16-B stores
  extern __shared__ __attribute__((aligned(16))) char smem[];
  constexpr int HALF = (NTK + 1) / 2 * 16;  // cols per half
  bf16* lds_t = (bf16*)smem + (size_t)wave * 16 * (HALF + 8);
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    const int n0 = h * ((NTK + 1) / 2);
    const int n1 = min(NTK, n0 + (NTK + 1) / 2);
    if (n0 >= NTK) break;
#pragma unroll
    for (int n = 0; n < NTK; ++n) {
      if (n >= n0 && n < n1) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row_l = (lane >> 4) * 4 + r;
          lds_t[(size_t)row_l * (HALF + 8) + (n - n0) * 16 + (lane & 15)] =
              f2bf(acc[n][r]);
        }
      }
    }
    // flush: 16 rows x (n1-n0)*16 cols
    const int cols = (n1 - n0) * 16;
    const int chunks = 16 * cols / 8;
    for (int c = lane; c < chunks; c += WAVE) {
      const int row_l = c / (cols / 8);
      const int col8 = (c % (cols / 8)) * 8;
      const long row = row0 + row_l;
      if (row < M) {
        const uint4 v =
            *(const uint4*)(lds_t + (size_t)row_l * (HALF + 8) + col8);
        *(uint4*)(dX + row * KP + n0 * 16 + col8) = v;
      }
    }
  }
}

extern "C" {

void launch_dgrad(const void* dZ, const void* W2, void* dX, long M, int KP,
                  int EP, hipStream_t stream) {
  const int NTK = KP / 16;
  const long grid = (M + 127) / 128;
  const int HALF = (NTK + 1) / 2 * 16;
  const int smem = 8 * 16 * (HALF + 8) * (int)sizeof(bf16);
#define DCASE(ntk)                                                           \
  case ntk:                                                                  \
    dgrad_kernel<ntk><<<grid, 512, smem, stream>>>(                          \
        (const bf16*)dZ, (const bf16*)W2, (bf16*)dX, M, EP);                 \
    break;
  switch (NTK) {
    DCASE(6) DCASE(8) DCASE(12) DCASE(16) DCASE(20) DCASE(24) DCASE(28)
    DCASE(32)
    default:
      printf("dgrad: unsupported KP=%d\n", KP);
  }
#undef DCASE
}

}  // extern "C"
