// K14 dgrad v2: dx[M, KP] = dz[M, EP=128] @ W2[KP, EP]^T — the combiner
// input gradient as a transposed-orientation MFMA streaming kernel
// (derived from head_fwd.hip; see PERF.md for the v1 story: reusing
// head_fwd directly was a wash with rocBLAS because its 64-batch tiles
// doubled the wave count at this 131-MB output).
//
// W2 is the [KP, EP] re-transpose of the stored [EP, KP] combiner weight
// (82 KB, transposed host-side per backward); with labels(=KP) as MFMA
// rows and dz rows as columns both fragments are contiguous 16-B runs.
// Block = 512 threads: tile [256 kp x 128 dz-rows] — 2x the batch width
// of head_fwd, so half the waves and per-wave overhead.  The dz tile
// (32 KB) stages block-cooperatively; the output transposes through a
// block-wide LDS patch (b64 writes, full 128-B line nontemporal stores).
// KP/EP tails fall back to rocBLAS in functional.py (KP%8==0, EP==128).

#include "common.h"

#define DG_LABS 256   // kp rows per block
#define DG_BATCH 128  // dz rows per block

__global__ __launch_bounds__(512) void dgrad2_kernel(
    const bf16* __restrict__ dz, const bf16* __restrict__ w2,
    bf16* __restrict__ dx, long M, long KP, int GYB) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int EP = 128;
  // XCD-aware swizzle (see head_fwd.hip)
  const int total = gridDim.x;
  const int lin = (total % 8 == 0)
      ? (int)(blockIdx.x % 8) * (total / 8) + (int)blockIdx.x / 8
      : (int)blockIdx.x;
  const int by = lin % GYB;
  const int bx = lin / GYB;
  const long mb0 = (long)by * DG_BATCH;
  const long lab0 = (long)bx * DG_LABS + wave * 32;
  const int kj = (lane >> 4) * 8;

  const long arow0 = lab0 + (lane & 15);
  const long arow1 = arow0 + 16;
  const bf16* ap0 = w2 + arow0 * EP + kj;
  const bf16* ap1 = w2 + arow1 * EP + kj;
  const bool a0ok = arow0 < KP;
  const bool a1ok = arow1 < KP;

  f32x4 acc[2][8];
#pragma unroll
  for (int rt = 0; rt < 2; ++rt)
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) acc[rt][nt] = f32x4{0.f, 0.f, 0.f, 0.f};
  const bf16x8 zero8 = {};

  // [128 dz rows][264]: conflict-free b128 fragment reads (see head_fwd)
  __shared__ bf16 smem2d[DG_BATCH][264];

  for (int t = threadIdx.x; t < DG_BATCH * 16; t += 512) {
    const int r = t >> 4;
    const int c = t & 15;
    const bf16x8 v = (mb0 + r < M)
        ? *(const bf16x8*)(dz + (mb0 + r) * EP + c * 8) : zero8;
    *(bf16x8*)&smem2d[r][c * 8] = v;
  }
  __syncthreads();

#define DG_LD(v, ok, p, kk) \
  const bf16x8 v = (ok) ? *(const bf16x8*)((p) + (kk)*32) : zero8;
#define DG_LB(v, nt, kk) \
  const bf16x8 v = \
      *(const bf16x8*)&smem2d[nt * 16 + (lane & 15)][(kk)*32 + kj];
#define DG_MM(rt, nt, kk)                                       \
  acc[rt][nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(        \
      a##rt##kk, b##nt, acc[rt][nt], 0, 0, 0);
  DG_LD(a00, a0ok, ap0, 0) DG_LD(a01, a0ok, ap0, 1)
  DG_LD(a02, a0ok, ap0, 2) DG_LD(a03, a0ok, ap0, 3)
  DG_LD(a10, a1ok, ap1, 0) DG_LD(a11, a1ok, ap1, 1)
  DG_LD(a12, a1ok, ap1, 2) DG_LD(a13, a1ok, ap1, 3)
#define DG_KSTEP(kk)                                                   \
  {                                                                    \
    DG_LB(b0, 0, kk) DG_LB(b1, 1, kk) DG_LB(b2, 2, kk)                \
    DG_LB(b3, 3, kk) DG_LB(b4, 4, kk) DG_LB(b5, 5, kk)                 \
    DG_LB(b6, 6, kk) DG_LB(b7, 7, kk)                                  \
    DG_MM(0, 0, kk) DG_MM(1, 0, kk) DG_MM(0, 1, kk) DG_MM(1, 1, kk)    \
    DG_MM(0, 2, kk) DG_MM(1, 2, kk) DG_MM(0, 3, kk) DG_MM(1, 3, kk)    \
    DG_MM(0, 4, kk) DG_MM(1, 4, kk) DG_MM(0, 5, kk) DG_MM(1, 5, kk)    \
    DG_MM(0, 6, kk) DG_MM(1, 6, kk) DG_MM(0, 7, kk) DG_MM(1, 7, kk)    \
  }
  DG_KSTEP(0) DG_KSTEP(1) DG_KSTEP(2) DG_KSTEP(3)
#undef DG_LD
#undef DG_LB
#undef DG_MM
#undef DG_KSTEP

  // dz tile is dead; reuse smem2d as the [128 rows][256+8 kp] patch
  __syncthreads();
  bf16 (*patch)[264] = smem2d;
#pragma unroll
  for (int rt = 0; rt < 2; ++rt) {
    const int labl = wave * 32 + rt * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int nt = 0; nt < 8; ++nt) {
      bf16x4 v4;
#pragma unroll
      for (int r = 0; r < 4; ++r) v4[r] = f2bf(acc[rt][nt][r]);
      *(bf16x4*)&patch[nt * 16 + (lane & 15)][labl] = v4;
    }
  }
  __syncthreads();

  // store: full 128-B line per 8 lanes per dz row
  const long cend = (KP - (long)bx * DG_LABS < DG_LABS)
                        ? KP - (long)bx * DG_LABS : DG_LABS;
  const int bcr = wave * 16 + (lane >> 3) * 2;  // 2 rows per lane pair set
#pragma unroll
  for (int rr = 0; rr < 2; ++rr) {
    const long grow = mb0 + bcr + rr;
#pragma unroll
    for (int p = 0; p < 4; ++p) {
      const int lo = p * 64 + (lane & 7) * 8;
      if (grow < M && lo < cend) {
        bf16* dst = dx + grow * KP + (long)bx * DG_LABS + lo;
        if (lo + 8 <= cend) {
          __builtin_nontemporal_store(
              *(const bf16x8*)&patch[bcr + rr][lo], (bf16x8*)dst);
        } else {
          for (int j = 0; j < cend - lo; ++j)
            dst[j] = patch[bcr + rr][lo + j];
        }
      }
    }
  }
}

extern "C" {

void launch_dgrad2(const void* dz, const void* w2, void* dx, long M, long KP,
                   hipStream_t stream) {
  const int GXL = (int)((KP + DG_LABS - 1) / DG_LABS);
  const int GYB = (int)((M + DG_BATCH - 1) / DG_BATCH);
  dgrad2_kernel<<<(long)GXL * GYB, 512, 0, stream>>>(
      (const bf16*)dz, (const bf16*)w2, (bf16*)dx, M, KP, GYB);
}

}  // extern "C"
