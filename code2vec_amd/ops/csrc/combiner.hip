// K3-K6: fused context combiner — MFMA GEMM (z = X @ W) with a
// LayerNorm + tanh + dropout epilogue, plus the fused elementwise/row-reduce
// part of its backward (dz from dOut).  The plain dgrad/wgrad GEMMs of the
// backward go through rocBLAS (ops/functional.py).
//
// Reference math: model/model.py:54-61 (Linear no-bias -> LayerNorm(E) ->
// tanh -> Dropout).  This kernel is MI355X-native: 64-lane waves,
// mfma_f32_16x16x32_bf16 tiles, fp32 accumulation, LDS-staged B operand,
// in-register row reductions for the LN statistics.
//
// Layouts: X [M, KP] bf16 row-major (or gathered on the fly from the
// embedding tables in the fused GATHER=1 variant); W is stored TRANSPOSED
// [EP, KP] bf16 so the MFMA B fragment is memory-contiguous; z/out [M, EP]
// bf16; gamma/beta [EP] fp32 (pad rows/cols zero); valid cols = E.
// Pad rows of W are zero => z pad cols are exactly zero => LN sums over all
// EP cols equal sums over the E valid cols.

#include "common.h"

#define LN_EPS 1e-5f
#define COMBINER_BWD_MAX_GRID 1024

// ---------------------------------------------------------------------------
// Forward.  Block = 256 threads = 4 waves; BM = 128 rows (wave w owns rows
// w*32..w*32+31 as two 16-row MFMA tiles); full EP width per block.
//
// Both MFMA operands load straight from memory — no LDS staging, no
// barriers: A fragments are contiguous 16-B pieces of X rows (HBM, consumed
// once); B fragments are contiguous 16-B pieces of the TRANSPOSED weight
// Wt [EP, KP] (<=240 KB, L2-resident, re-read per block).  The transposed
// parameter layout exists exactly so the B fragment (8 consecutive k at one
// output column) is memory-contiguous.
//
// MFMA fragment maps (gfx950 mfma_f32_16x16x32_bf16):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + j], j = 0..7
//   B: lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C/D: lane l, reg r holds C[row = (l>>4)*4 + r][col = l&15]
// GATHER=1 fuses K1/K2 into the GEMM: the A fragment reads the embedding
// tables directly (segment-resolved), never materializing the concat
// tensor.  Segment boundaries are 8-element multiples, so a 16-B fragment
// piece never straddles one.
// MT: 16-row M tiles per wave.  MINW: min waves/SIMD the register allocator
// must honor (occupancy experiment C2V_COMBINER_OCC — the default NT=8/MT=2
// build lands at 230 VGPR+AGPR = 2 waves/SIMD; forcing 3 trades ILP/spill
// for latency hiding on the TA-issue-bound A-fragment loads).
template <int NT, int EPI, int GATHER, int MT = 2, int MINW = 1>
__global__ __launch_bounds__(256, MINW) void combiner_fwd_kernel(
    const bf16* __restrict__ X, const bf16* __restrict__ Wt,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    bf16* __restrict__ out, bf16* __restrict__ z_save,
    float* __restrict__ mean_save, float* __restrict__ rstd_save,
    long M, int KP, int E, float p, float inv1mp,
    unsigned long long seed, const unsigned long long* __restrict__ rng_off,
    const int* __restrict__ g_starts, const int* __restrict__ g_paths,
    const int* __restrict__ g_ends, const bf16* __restrict__ g_term,
    const bf16* __restrict__ g_path, int TS, int PS) {
  const int EP = NT * 16;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const long row0 = (long)blockIdx.x * (MT * 64);
  // RNG offset lives in DEVICE memory so hipGraph replays of a captured
  // training step advance the dropout stream (a by-value argument would
  // freeze the mask at capture time); bumped by bump_u64_kernel per call
  const unsigned long long offset = p > 0.0f ? rng_off[0] : 0ull;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* lds_gamma = (float*)smem;
  float* lds_beta = lds_gamma + EP;  // gamma/beta live for the whole kernel
  for (int c = threadIdx.x; c < EP; c += blockDim.x) {
    lds_gamma[c] = gamma[c];
    lds_beta[c] = beta[c];
  }
  __syncthreads();

  const int NK = KP / 32;
  f32x4 acc[MT][NT];
#pragma unroll
  for (int mi = 0; mi < MT; ++mi)
#pragma unroll
    for (int n = 0; n < NT; ++n) acc[mi][n] = f32x4{0.f, 0.f, 0.f, 0.f};

  long arow[MT];
#pragma unroll
  for (int mi = 0; mi < MT; ++mi) {
    long r = row0 + wave * (MT * 16) + mi * 16 + (lane & 15);
    arow[mi] = r < M ? r : (M - 1);
  }
  // fused-gather mode: this lane's three embedding rows, loaded once
  const bf16* seg_base[MT][3];
  if (GATHER) {
#pragma unroll
    for (int mi = 0; mi < MT; ++mi) {
      seg_base[mi][0] = g_term + (size_t)g_starts[arow[mi]] * TS;
      seg_base[mi][1] = g_path + (size_t)g_paths[arow[mi]] * PS;
      seg_base[mi][2] = g_term + (size_t)g_ends[arow[mi]] * TS;
    }
  }
  const int kj = (lane >> 4) * 8;  // this lane's k sub-offset within a K-step

  // B tile staged in LDS, double-buffered, T14 split (load regs early,
  // ds_write after MFMA).  Image: [e][k8] with e-stride 40 elements (80 B)
  // and k in the low position — both the 16-B ds_writes (consecutive
  // (e,k8) chunks) and the per-lane fragment ds_reads spread banks with
  // no conflicts.  Fragment (n, lane): e = n*16 + (lane&15), k8 = lane>>4.
  constexpr int ESTRIDE = 40;  // elements per e column (32 data + 8 pad)
  bf16* lds_b = (bf16*)(smem + 2 * EP * sizeof(float));
  const size_t bufsz = (size_t)EP * ESTRIDE;
  constexpr int CPT = (NT + 3) / 4;  // 16-B chunks per thread per K-step
  const int schunks = 4 * EP;  // (EP cols) x (32/8 k-groups)
  const int s_e = threadIdx.x >> 2;
  const int s_k8 = threadIdx.x & 3;

  // direct global->LDS staging of K-tile kk into buffer buf (short-lived
  // register round trip only; an address-taken staging array gets demoted
  // to scratch INSIDE the loop by hipcc)
#define STAGE(kk, buf)                                                        \
  _Pragma("unroll") for (int i_ = 0; i_ < CPT; ++i_) {                        \
    const int c_ = (int)threadIdx.x + i_ * 256;                               \
    if (c_ < schunks)                                                         \
      *(uint4*)(lds_b + (buf) * bufsz + (size_t)(s_e + i_ * 64) * ESTRIDE +   \
                s_k8 * 8) =                                                   \
          *(const uint4*)(Wt + (size_t)(s_e + i_ * 64) * KP + (kk) * 32 +     \
                          s_k8 * 8);                                          \
  }

  STAGE(0, 0)
  __syncthreads();

  for (int kk = 0; kk < NK; ++kk) {
    const int buf = kk & 1;
    if (kk + 1 < NK) STAGE(kk + 1, buf ^ 1)
    bf16x8 a[MT];
    const int kfrag = kk * 32 + kj;
#pragma unroll
    for (int mi = 0; mi < MT; ++mi) {
      if (GATHER) {
        bf16x8 v = {};
        if (kfrag < TS)
          v = *(const bf16x8*)(seg_base[mi][0] + kfrag);
        else if (kfrag < TS + PS)
          v = *(const bf16x8*)(seg_base[mi][1] + kfrag - TS);
        else if (kfrag < 2 * TS + PS)
          v = *(const bf16x8*)(seg_base[mi][2] + kfrag - TS - PS);
        a[mi] = v;
      } else {
        a[mi] = *(const bf16x8*)(X + arow[mi] * KP + kfrag);
      }
    }
    // explicit 2-deep B pipeline: read fragment n+1 while n's MFMAs issue
    const bf16* bbase =
        lds_b + buf * bufsz + (size_t)(lane & 15) * ESTRIDE + (lane >> 4) * 8;
    bf16x8 b_cur = *(const bf16x8*)(bbase);
    bf16x8 b_nxt;
#pragma unroll
    for (int n = 0; n < NT; ++n) {
      if (n + 1 < NT)
        b_nxt = *(const bf16x8*)(bbase + (size_t)(n + 1) * 16 * ESTRIDE);
#pragma unroll
      for (int mi = 0; mi < MT; ++mi)
        acc[mi][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[mi], b_cur, acc[mi][n], 0, 0, 0);
      b_cur = b_nxt;
    }
    __syncthreads();
  }
#undef STAGE

  // ---- epilogue: LayerNorm(E) + tanh + dropout ----
  // EPI==1: values staged per-wave in LDS (padded row stride) and flushed
  // as contiguous 16-B chunks — the wave's 32 output rows are contiguous
  // in memory, so the flush is one coalesced 8-KB span per tensor.
  const float invE = 1.0f / (float)E;
  // The bounce tile ALIASES the B staging buffer: the K-loop's final
  // __syncthreads() guarantees every wave is done reading lds_b, so the
  // epilogue reuses its space — block LDS drops from 56.3 KB to 35.8 KB
  // (EP=128) and LDS stops capping occupancy at 2 blocks/CU.
  bf16* lds_t = nullptr;
  if (EPI == 1) {
    lds_t = (bf16*)(smem + 2 * EP * sizeof(float)) +
            (size_t)wave * (MT * 16) * (EP + 8);
  }
  float mean_r[MT][4], rstd_r[MT][4];
#pragma unroll
  for (int mi = 0; mi < MT; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long row = row0 + wave * (MT * 16) + mi * 16 + (lane >> 4) * 4 + r;
      float s1 = 0.f, s2 = 0.f;
#pragma unroll
      for (int n = 0; n < NT; ++n) {
        const float v = acc[mi][n][r];
        s1 += v;
        s2 += v * v;
      }
      s1 = group16_reduce_sum(s1);
      s2 = group16_reduce_sum(s2);
      const float mean = s1 * invE;
      const float var = fmaxf(s2 * invE - mean * mean, 0.0f);
      const float rstd = rsqrtf(var + LN_EPS);
      mean_r[mi][r] = mean;
      rstd_r[mi][r] = rstd;
      if (row < M && (lane & 15) == 0) {
        mean_save[row] = mean;
        rstd_save[row] = rstd;
      }
    }
  }

  // two value passes: tensor 0 = z (raw GEMM out), tensor 1 = out
  // (LN+tanh+dropout); each pass stages (EPI==1) or stores scalar (EPI==0).
#pragma unroll
  for (int pass = 0; pass < 2; ++pass) {
    bf16* dst_base = pass == 0 ? z_save : out;
#pragma unroll
    for (int mi = 0; mi < MT; ++mi) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row_l = mi * 16 + (lane >> 4) * 4 + r;
        const long row = row0 + wave * (MT * 16) + row_l;
        const float mean = mean_r[mi][r];
        const float rstd = rstd_r[mi][r];
        if (EPI == 0 && row >= M) continue;
#pragma unroll
        for (int n = 0; n < NT; ++n) {
          const int col = n * 16 + (lane & 15);
          const float zv = acc[mi][n][r];
          float val = 0.f;
          if (col < E) {
            if (pass == 0) {
              val = zv;
            } else {
              const float xhat = (zv - mean) * rstd;
              const float u = xhat * lds_gamma[col] + lds_beta[col];
              float y = fast_tanh(u);
              if (p > 0.0f) {
                const float u01 = rng_uniform(
                    seed, offset + (unsigned long long)row * EP + col);
                y = (u01 >= p) ? y * inv1mp : 0.0f;
              }
              val = y;
            }
          }
          if (EPI == 1) {
            lds_t[(size_t)row_l * (EP + 8) + col] = f2bf(val);
          } else {
            dst_base[row * EP + col] = f2bf(val);
          }
        }
      }
    }
    if (EPI == 1) {
      // flush the wave's [MT*16, EP] tile as 16-B chunks (coalesced)
      const int chunks = MT * 16 * EP / 8;
      for (int c = lane; c < chunks; c += WAVE) {
        const int row_l = c / (EP / 8);
        const int col8 = (c % (EP / 8)) * 8;
        const long row = row0 + wave * (MT * 16) + row_l;
        if (row < M) {
          const uint4 v =
              *(const uint4*)(lds_t + (size_t)row_l * (EP + 8) + col8);
          *(uint4*)(dst_base + row * EP + col8) = v;
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Backward (elementwise + row-reduce part): given dOut, produce
// dz = LN/tanh/dropout chain; accumulate dgamma/dbeta.
//
// Key trick: the forward OUTPUT (post-dropout tanh) is already kept alive by
// the attention autograd node, so backward recovers both the dropout mask
// (out != 0 => kept) and the tanh value (y = out*(1-p)) from it — no tanh
// and no RNG recompute.  (A kept element whose tanh rounds to bf16 zero is
// treated as dropped; |u| would have to be < ~1e-38 — measure-zero.)
//
// One wave per row, single pass: each lane owns column pairs (2 bf16 = 4 B
// vector loads); dgamma/dbeta accumulate per-lane across the grid-stride
// loop and flush with one atomic per lane at kernel end.
template <int NPAIR>  // ceil(EP/128) column-pair iterations per lane
__global__ __launch_bounds__(256) void combiner_bwd_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ z,
    const bf16* __restrict__ out, const float* __restrict__ mean,
    const float* __restrict__ rstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, bf16* __restrict__ dz,
    float* __restrict__ dgamma_part, float* __restrict__ dbeta_part, long M,
    int EP, int E, float p, float inv1mp) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  const float invE = 1.0f / (float)E;
  const float one_mp = 1.0f - p;  // out -> y scale (1 when p == 0)

  float g_c[NPAIR][2];
  float acc_dg[NPAIR][2] = {};
  float acc_db[NPAIR][2] = {};
  int cols[NPAIR];
#pragma unroll
  for (int i = 0; i < NPAIR; ++i) {
    cols[i] = i * 128 + lane * 2;
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int c = cols[i] + j;
      g_c[i][j] = c < EP ? gamma[c] : 0.f;
    }
  }

  long row = (long)blockIdx.x * waves_per_block + wave;
  const long stride = (long)gridDim.x * waves_per_block;
  for (; row < M; row += stride) {
    const float mu = mean[row];
    const float rs = rstd[row];
    const bf16* drow = dout + row * EP;
    const bf16* zrow = z + row * EP;
    const bf16* orow = out + row * EP;
    bf16* dzrow = dz + row * EP;
    float xhat[NPAIR][2], du[NPAIR][2];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int i = 0; i < NPAIR; ++i) {
      const int c0 = cols[i];
      bf16x2 zv2 = {bf16(0.f), bf16(0.f)}, dy2 = {bf16(0.f), bf16(0.f)};
      bf16x2 ov2 = {bf16(0.f), bf16(0.f)};
      if (c0 < EP) {
        zv2 = *(const bf16x2*)(zrow + c0);
        dy2 = *(const bf16x2*)(drow + c0);
        ov2 = *(const bf16x2*)(orow + c0);
      }
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const float xh = (bf2f(zv2[j]) - mu) * rs;
        const float ov = bf2f(ov2[j]);
        // kept => y = out*(1-p), dy_eff = dout/(1-p); dropped/pad => 0
        const float y = ov * one_mp;
        const float dy = (ov != 0.f) ? bf2f(dy2[j]) * inv1mp : 0.f;
        const float d = dy * (1.0f - y * y);
        xhat[i][j] = xh;
        du[i][j] = d;
        acc_dg[i][j] += d * xh;
        acc_db[i][j] += d;
        const float h = d * g_c[i][j];
        s1 += h;
        s2 += h * xh;
      }
    }
    s1 = wave_reduce_sum(s1) * invE;
    s2 = wave_reduce_sum(s2) * invE;
#pragma unroll
    for (int i = 0; i < NPAIR; ++i) {
      const int c0 = cols[i];
      if (c0 >= EP) continue;
      bf16x2 o;
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const bool valid = (c0 + j) < E;
        const float h = du[i][j] * g_c[i][j];
        o[j] = f2bf(valid ? rs * (h - s1 - xhat[i][j] * s2) : 0.f);
      }
      *(bf16x2*)(dzrow + c0) = o;
    }
  }
  // block-combine the per-lane dgamma/dbeta partials in LDS, then one
  // partials-row store per block (summed host-side; avoids the
  // all-waves-hit-128-addresses atomic serialization).  Each column is
  // owned by exactly ONE lane within a wave, so every wave writes its own
  // row and the final 4-way sum runs in fixed order — the earlier LDS
  // atomicAdd combine was ordering-nondeterministic at the fp32 ulp and
  // made identical training runs diverge (see PERF.md determinism note).
  __shared__ float red_g[4][128 * NPAIR];
  __shared__ float red_b[4][128 * NPAIR];
#pragma unroll
  for (int i = 0; i < NPAIR; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int c = cols[i] + j;
      if (c < 128 * NPAIR) {
        red_g[wave][c] = (c < EP) ? acc_dg[i][j] : 0.f;
        red_b[wave][c] = (c < EP) ? acc_db[i][j] : 0.f;
      }
    }
  __syncthreads();
  for (int c = threadIdx.x; c < EP; c += blockDim.x) {
    dgamma_part[(long)blockIdx.x * EP + c] =
        red_g[0][c] + red_g[1][c] + red_g[2][c] + red_g[3][c];
    dbeta_part[(long)blockIdx.x * EP + c] =
        red_b[0][c] + red_b[1][c] + red_b[2][c] + red_b[3][c];
  }
}

// V2 of the backward for the flagship EP=128 shape: TWO rows per wave
// (half-wave each) so every load/store is 8 B (bf16x4) instead of 4 B —
// halves the vector-memory op count of this latency-bound row sweep.
// The LN reductions shrink to 32-lane xor trees (offsets 16..1 stay
// within a half-wave); dgamma/dbeta halves combine with ONE fixed
// cross-half shuffle before the per-wave-row block reduce, keeping the
// bitwise run-to-run determinism of the fixed-order combine.
__device__ __forceinline__ float half_wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 16; off > 0; off >>= 1) x += __shfl_xor(x, off);
  return x;
}

__global__ __launch_bounds__(256) void combiner_bwd_v2_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ z,
    const bf16* __restrict__ out, const float* __restrict__ mean,
    const float* __restrict__ rstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, bf16* __restrict__ dz,
    float* __restrict__ dgamma_part, float* __restrict__ dbeta_part, long M,
    int E, float p, float inv1mp) {
  const int EP = 128;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int half = lane >> 5;
  const int c0 = (lane & 31) * 4;
  const float invE = 1.0f / (float)E;
  const float one_mp = 1.0f - p;

  float g_c[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) g_c[j] = gamma[c0 + j];
  float acc_dg[4] = {}, acc_db[4] = {};

  const long wpb = blockDim.x / WAVE;           // wave-pairs per block
  long pr = (long)blockIdx.x * wpb + wave;      // row-pair index
  const long pstride = (long)gridDim.x * wpb;
  const long npairs = (M + 1) >> 1;
  for (; pr < npairs; pr += pstride) {
    const long row = 2 * pr + half;
    const bool rok = row < M;
    const float mu = rok ? mean[row] : 0.f;
    const float rs = rok ? rstd[row] : 0.f;
    bf16x4 zv = {}, dy4 = {}, ov4 = {};
    if (rok) {
      zv = *(const bf16x4*)(z + row * EP + c0);
      dy4 = *(const bf16x4*)(dout + row * EP + c0);
      ov4 = *(const bf16x4*)(out + row * EP + c0);
    }
    float xhat[4], du[4];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const float xh = (bf2f(zv[j]) - mu) * rs;
      const float ov = bf2f(ov4[j]);
      const float y = ov * one_mp;
      const float dy = (ov != 0.f) ? bf2f(dy4[j]) * inv1mp : 0.f;
      const float d = dy * (1.0f - y * y);
      xhat[j] = xh;
      du[j] = d;
      acc_dg[j] += d * xh;
      acc_db[j] += d;
      const float h = d * g_c[j];
      s1 += h;
      s2 += h * xh;
    }
    s1 = half_wave_reduce_sum(s1) * invE;
    s2 = half_wave_reduce_sum(s2) * invE;
    if (rok) {
      bf16x4 o;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const bool valid = (c0 + j) < E;
        const float h = du[j] * g_c[j];
        o[j] = f2bf(valid ? rs * (h - s1 - xhat[j] * s2) : 0.f);
      }
      *(bf16x4*)(dz + row * EP + c0) = o;
    }
  }

  // fixed-order cross-half combine: half 0's lane adds half 1's partial
  __shared__ float red_g[4][128];
  __shared__ float red_b[4][128];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const float og = __shfl_xor(acc_dg[j], 32);
    const float ob = __shfl_xor(acc_db[j], 32);
    if (half == 0) {
      const int c = c0 + j;
      red_g[wave][c] = (c < EP) ? acc_dg[j] + og : 0.f;
      red_b[wave][c] = (c < EP) ? acc_db[j] + ob : 0.f;
    }
  }
  __syncthreads();
  for (int c = threadIdx.x; c < EP; c += blockDim.x) {
    dgamma_part[(long)blockIdx.x * EP + c] =
        red_g[0][c] + red_g[1][c] + red_g[2][c] + red_g[3][c];
    dbeta_part[(long)blockIdx.x * EP + c] =
        red_b[0][c] + red_b[1][c] + red_b[2][c] + red_b[3][c];
  }
}

__global__ void bump_u64_kernel(unsigned long long* p,
                                unsigned long long delta) {
  if (threadIdx.x == 0 && blockIdx.x == 0) p[0] += delta;
}

extern "C" {

void launch_bump_u64(void* p, unsigned long long delta,
                     hipStream_t stream) {
  bump_u64_kernel<<<1, 1, 0, stream>>>((unsigned long long*)p, delta);
}

void launch_combiner_fwd_impl(const void* X, const void* W,
                         const float* gamma,
                         const float* beta, void* out, void* z, float* mean,
                         float* rstd, long M, int KP, int EP, int E, float p,
                         unsigned long long seed,
                         const unsigned long long* rng_off,
                         int epilogue_mode, int gather, const int* starts,
                         const int* paths, const int* ends, const void* term,
                         const void* path, int TS, int PS,
                         hipStream_t stream) {
  const int NT = EP / 16;
  const char* mt_env = getenv("C2V_COMBINER_MT");
  const int mt = (mt_env && mt_env[0] == '1') ? 1 : 2;
  // Occupancy-forced register allocation (C2V_COMBINER_OCC=0/3/4): the
  // unconstrained build burns 230 VGPR+AGPR -> 2 waves/SIMD; with the
  // epilogue bounce aliased over the dead B buffer, LDS fits 4 blocks/CU
  // and __launch_bounds__(256,4) repacks to 128 VGPR (6 spills) ->
  // measured 116.7 / 97.1 / 92.3 us (occ 2/3/4) at top11 — default 4.
  // (guarded to NT<=10 in the dispatch: larger encode widths spill >1 KB
  // scratch/lane under the cap and must keep the unconstrained build)
  const char* occ_env = getenv("C2V_COMBINER_OCC");
  const int occ = occ_env ? atoi(occ_env) : 4;
  const long grid = (M + mt * 64 - 1) / (mt * 64);
  const float inv1mp = p > 0.f ? 1.0f / (1.0f - p) : 1.0f;
  int epi = epilogue_mode;
  (void)gather;
  const int bbuf = 2 * EP * 40 * (int)sizeof(bf16);
  int smem = 2 * EP * sizeof(float) + bbuf;
  if (epi == 1) {
    // bounce aliases the dead B buffer; allocate the max of the two
    const int bounce = 4 * mt * 16 * (EP + 8) * (int)sizeof(bf16);
    const int need = 2 * EP * (int)sizeof(float) + (bounce > bbuf ? bounce : bbuf);
    if (need <= 160 * 1024 - 2048) smem = need;
    else epi = 0;
  }
#define CASE(nt)                                                              \
  case nt:                                                                    \
    if (gather && mt == 1)                                                    \
      combiner_fwd_kernel<nt, 1, 1, 1><<<grid, 256, smem, stream>>>(          \
          nullptr, (const bf16*)W, gamma, beta, (bf16*)out, (bf16*)z,         \
          mean, rstd, M, KP, E, p, inv1mp, seed, rng_off, starts, paths,      \
          ends, (const bf16*)term, (const bf16*)path, TS, PS);                \
    else if (gather)                                                          \
      combiner_fwd_kernel<nt, 1, 1, 2><<<grid, 256, smem, stream>>>(          \
          nullptr, (const bf16*)W, gamma, beta, (bf16*)out, (bf16*)z,         \
          mean, rstd, M, KP, E, p, inv1mp, seed, rng_off, starts, paths,      \
          ends, (const bf16*)term, (const bf16*)path, TS, PS);                \
    else if (epi == 1 && occ == 4 && nt <= 10)                                \
      combiner_fwd_kernel<nt, 1, 0, 2, 4><<<grid, 256, smem, stream>>>(       \
          (const bf16*)X, (const bf16*)W, gamma, beta, (bf16*)out, (bf16*)z,  \
          mean, rstd, M, KP, E, p, inv1mp, seed, rng_off, nullptr, nullptr,   \
          nullptr, nullptr, nullptr, 0, 0);                                   \
    else if (epi == 1 && occ == 3 && nt <= 10)                                \
      combiner_fwd_kernel<nt, 1, 0, 2, 3><<<grid, 256, smem, stream>>>(       \
          (const bf16*)X, (const bf16*)W, gamma, beta, (bf16*)out, (bf16*)z,  \
          mean, rstd, M, KP, E, p, inv1mp, seed, rng_off, nullptr, nullptr,   \
          nullptr, nullptr, nullptr, 0, 0);                                   \
    else if (epi == 1)                                                        \
      combiner_fwd_kernel<nt, 1, 0><<<grid, 256, smem, stream>>>(             \
          (const bf16*)X, (const bf16*)W, gamma, beta, (bf16*)out, (bf16*)z,  \
          mean, rstd, M, KP, E, p, inv1mp, seed, rng_off, nullptr, nullptr,   \
          nullptr, nullptr, nullptr, 0, 0);                                   \
    else                                                                      \
      combiner_fwd_kernel<nt, 0, 0><<<grid, 256, smem, stream>>>(             \
          (const bf16*)X, (const bf16*)W, gamma, beta, (bf16*)out, (bf16*)z,  \
          mean, rstd, M, KP, E, p, inv1mp, seed, rng_off, nullptr, nullptr,   \
          nullptr, nullptr, nullptr, 0, 0);                                   \
    break;
  switch (NT) {
    CASE(2) CASE(4) CASE(6) CASE(8) CASE(10)
    CASE(12) CASE(14) CASE(16) CASE(18) CASE(20)
    default:
      printf("combiner_fwd: unsupported EP=%d\n", EP);
  }
#undef CASE
}

void launch_combiner_fwd(const void* X, const void* W, const float* gamma,
                         const float* beta, void* out, void* z, float* mean,
                         float* rstd, long M, int KP, int EP, int E, float p,
                         unsigned long long seed,
                         const unsigned long long* rng_off,
                         int epilogue_mode, hipStream_t stream) {
  launch_combiner_fwd_impl(X, W, gamma, beta, out, z, mean, rstd, M, KP, EP,
                           E, p, seed, rng_off, epilogue_mode, 0, nullptr,
                           nullptr, nullptr, nullptr, nullptr, 0, 0, stream);
}

void launch_gather_combiner_fwd(const int* starts, const int* paths,
                                const int* ends, const void* term,
                                const void* path, int TS, int PS,
                                const void* W, const float* gamma,
                                const float* beta, void* out, void* z,
                                float* mean, float* rstd, long M, int KP,
                                int EP, int E, float p,
                                unsigned long long seed,
                                const unsigned long long* rng_off,
                                hipStream_t stream) {
  launch_combiner_fwd_impl(nullptr, W, gamma, beta, out, z, mean, rstd, M,
                           KP, EP, E, p, seed, rng_off, 1, 1, starts, paths,
                           ends, term, path, TS, PS, stream);
}

void launch_combiner_bwd(const void* dout, const void* z, const void* out,
                         const float* mean, const float* rstd,
                         const float* gamma, const float* beta, void* dz,
                         float* dgamma, float* dbeta, long M, int EP, int E,
                         float p, hipStream_t stream) {
  const int block = 256;
  const int wpb = block / WAVE;
  const int grid = (int)min((M + wpb - 1) / wpb, (long)COMBINER_BWD_MAX_GRID);
  const float inv1mp = p > 0.f ? 1.0f / (1.0f - p) : 1.0f;
  const int npair = (EP + 127) / 128;
  // V2 (two rows per wave, 8-B vector ops) for the EP=128 flagship shape;
  // C2V_CB_V2=0 reverts to the one-row-per-wave kernel
  static const char* v2e = getenv("C2V_CB_V2");
  if (EP == 128 && (v2e == nullptr || v2e[0] != '0')) {
    // SAME grid as v1: the host sums dgamma/dbeta partials over this many
    // block rows, so every block must write its row (idle pairs write 0)
    combiner_bwd_v2_kernel<<<grid, block, 0, stream>>>(
        (const bf16*)dout, (const bf16*)z, (const bf16*)out, mean, rstd,
        gamma, beta, (bf16*)dz, dgamma, dbeta, M, E, p, inv1mp);
    return;
  }
#define BCASE(np)                                                             \
  case np:                                                                    \
    combiner_bwd_kernel<np><<<grid, block, 0, stream>>>(                      \
        (const bf16*)dout, (const bf16*)z, (const bf16*)out, mean, rstd,      \
        gamma, beta, (bf16*)dz, dgamma, dbeta, M, EP, E, p, inv1mp);          \
    break;
  switch (npair) {
    BCASE(1) BCASE(2) BCASE(3) BCASE(4)
    default:
      printf("combiner_bwd: unsupported EP=%d\n", EP);
  }
#undef BCASE
}

}  // extern "C"
