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
  // fragment images, 80-B column stride: a_img [128 l][40], b_img [EP][40]
  constexpr int ESTRIDE = 40;
  bf16* a_img = (bf16*)smem;                       // 128*40*2 = 10240 B
  bf16* b_img = a_img + 128 * ESTRIDE;             // EP*40*2

  f32x4 acc[NT];
#pragma unroll
  for (int n = 0; n < NT; ++n) acc[n] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kj = (lane >> 4) * 8;
  const int NK = (int)(B / 32);

  for (int kk = 0; kk < NK; ++kk) {
    __syncthreads();  // previous iteration's reads done
    // stage dlogits^T tile: threads read dlogits[b][l0 + c8*8 .. +8]
    {
      const int chunks = 32 * 128 / 8;  // 512: exactly one per thread? 512thr
      const int c = threadIdx.x;
      if (c < chunks) {
        const int brow = c / 16;         // 0..31
        const int l8 = (c % 16) * 8;     // 0..120
        const long b = (long)kk * 32 + brow;
        bf16 v[8];
        uint4 vv = {0, 0, 0, 0};
        if (l0 + l8 + 8 <= L)
          vv = *(const uint4*)(dlogits + b * L + l0 + l8);
        *(uint4*)v = vv;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          a_img[(size_t)(l8 + j) * ESTRIDE + brow] = v[j];
      }
    }
    // stage cv tile: threads read cv[b][e8*8 .. +8]
    {
      const int chunks = 32 * EP / 8;
      for (int c = threadIdx.x; c < chunks; c += blockDim.x) {
        const int brow = c / (EP / 8);
        const int e8 = (c % (EP / 8)) * 8;
        const long b = (long)kk * 32 + brow;
        bf16 v[8];
        *(uint4*)v = *(const uint4*)(cv + b * EP + e8);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          b_img[(size_t)(e8 + j) * ESTRIDE + brow] = v[j];
      }
    }
    __syncthreads();
    // wave w owns L rows [w*16, w*16+16)
    const bf16x8 a =
        *(const bf16x8*)(a_img + (size_t)(wave * 16 + (lane & 15)) * ESTRIDE + kj);
#pragma unroll
    for (int n = 0; n < NT; ++n) {
      const bf16x8 b = *(const bf16x8*)(
          b_img + (size_t)(n * 16 + (lane & 15)) * ESTRIDE + kj);
      acc[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[n], 0, 0, 0);
    }
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
  const int smem = (128 + EP) * 40 * (int)sizeof(bf16);
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
