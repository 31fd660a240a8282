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
// Layouts: X [M, KP] bf16 row-major; W [KP, EP] bf16 row-major (B-operand);
// z/out [M, EP] bf16; gamma/beta [EP] fp32 (pad cols zero); valid cols = E.
// Pad cols of W are zero => z pad cols are exactly zero => LN sums over all
// EP cols equal sums over the E valid cols.

#include "common.h"

#define LN_EPS 1e-5f

// ---------------------------------------------------------------------------
// Forward.  Block = 256 threads = 4 waves; BM = 128 rows (wave w owns rows
// w*32..w*32+31 as two 16-row MFMA tiles); full EP width per block.
// A fragments load straight from global (16-B per lane); B tiles are staged
// per K-step into an LDS image shaped for conflict-free ds_read_b128.
//
// MFMA fragment maps (gfx950 mfma_f32_16x16x32_bf16):
//   A: lane l holds A[row = l&15][k = (l>>4)*8 + j], j = 0..7
//   B: lane l holds B[k = (l>>4)*8 + j][col = l&15]
//   C/D: lane l, reg r holds C[row = (l>>4)*4 + r][col = l&15]
template <int NT>  // NT = EP/16 column tiles
__global__ __launch_bounds__(256) void combiner_fwd_kernel(
    const bf16* __restrict__ X, const bf16* __restrict__ W,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    bf16* __restrict__ out, bf16* __restrict__ z_save,
    float* __restrict__ mean_save, float* __restrict__ rstd_save,
    long M, int KP, int E, float p, float inv1mp,
    unsigned long long seed, unsigned long long offset) {
  const int EP = NT * 16;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const long row0 = (long)blockIdx.x * 128;

  extern __shared__ __attribute__((aligned(16))) char smem[];
  // double-buffered B image: [2][NT][64 lanes][8 bf16]
  bf16* lds_b = (bf16*)smem;
  const int bbytes = 2 * NT * 64 * 8 * sizeof(bf16);
  float* lds_gamma = (float*)(smem + bbytes);
  float* lds_beta = lds_gamma + EP;

  for (int c = threadIdx.x; c < EP; c += blockDim.x) {
    lds_gamma[c] = gamma[c];
    lds_beta[c] = beta[c];
  }

  const int NK = KP / 32;
  f32x4 acc[2][NT];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int n = 0; n < NT; ++n) acc[mi][n] = f32x4{0.f, 0.f, 0.f, 0.f};

  // A row this lane reads for tile mi: row0 + wave*32 + mi*16 + (lane&15)
  long arow[2];
  bool arow_ok[2];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
    long r = row0 + wave * 32 + mi * 16 + (lane & 15);
    arow_ok[mi] = r < M;
    arow[mi] = arow_ok[mi] ? r : (M - 1);
  }
  const int kj = (lane >> 4) * 8;  // this lane's k sub-offset within a K-step

  // stage B K-step kk into buffer buf
  auto stage_b = [&](int kk, int buf) {
    const int chunks = 32 * EP / 8;  // 16-B chunks in a [32, EP] W tile
    for (int c = threadIdx.x; c < chunks; c += blockDim.x) {
      const int krow = c / (EP / 8);
      const int col8 = (c % (EP / 8)) * 8;
      const bf16* src = W + (long)(kk * 32 + krow) * EP + col8;
      bf16 vals[8];
      *(uint4*)vals = *(const uint4*)src;
      const int n = col8 / 16;
      const int base_l = (col8 & 15) + (krow >> 3) * 16;
      const int jslot = krow & 7;
      bf16* dst = lds_b + (((long)buf * NT + n) * 64) * 8;
#pragma unroll
      for (int j = 0; j < 8; ++j) dst[(base_l + j) * 8 + jslot] = vals[j];
    }
  };

  stage_b(0, 0);
  __syncthreads();

  for (int kk = 0; kk < NK; ++kk) {
    const int buf = kk & 1;
    bf16x8 a[2];
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
      const bf16* ap = X + arow[mi] * KP + kk * 32 + kj;
      a[mi] = *(const bf16x8*)ap;
    }
#pragma unroll
    for (int n = 0; n < NT; ++n) {
      const bf16x8 b =
          *(const bf16x8*)(lds_b + (((long)buf * NT + n) * 64 + lane) * 8);
      acc[0][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[0], b, acc[0][n], 0, 0, 0);
      acc[1][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a[1], b, acc[1][n], 0, 0, 0);
    }
    if (kk + 1 < NK) {
      stage_b(kk + 1, buf ^ 1);
    }
    __syncthreads();
  }

  // ---- epilogue: LayerNorm(E) + tanh + dropout, in-register ----
  const float invE = 1.0f / (float)E;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long row = row0 + wave * 32 + mi * 16 + (lane >> 4) * 4 + r;
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
      const bool row_ok = row < M;
      if (row_ok && (lane & 15) == 0) {
        mean_save[row] = mean;
        rstd_save[row] = rstd;
      }
      if (!row_ok) continue;
      bf16* zrow = z_save + row * EP;
      bf16* orow = out + row * EP;
#pragma unroll
      for (int n = 0; n < NT; ++n) {
        const int col = n * 16 + (lane & 15);
        const float zv = acc[mi][n][r];
        float ov = 0.f;
        float zw = 0.f;
        if (col < E) {
          zw = zv;
          const float xhat = (zv - mean) * rstd;
          const float u = xhat * lds_gamma[col] + lds_beta[col];
          float y = tanhf(u);
          if (p > 0.0f) {
            const float u01 =
                rng_uniform(seed, offset + (unsigned long long)row * EP + col);
            y = (u01 >= p) ? y * inv1mp : 0.0f;
          }
          ov = y;
        }
        zrow[col] = f2bf(zw);
        orow[col] = f2bf(ov);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Backward (elementwise + row-reduce part): given dOut, produce
// dz = LN/tanh/dropout chain; accumulate dgamma/dbeta.
// One wave per row; two passes over the row's columns re-reading from cache
// (rows are 256-640 B -> L1-resident between passes).
__global__ void combiner_bwd_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ z,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    bf16* __restrict__ dz, float* __restrict__ dgamma_part,
    float* __restrict__ dbeta_part, long M, int EP, int E, float p,
    float inv1mp, unsigned long long seed, unsigned long long offset) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  const float invE = 1.0f / (float)E;

  // per-lane dgamma/dbeta accumulators for this lane's columns
  float acc_dg[8];  // EP/64 <= 8 (EP <= 512)
  float acc_db[8];
  const int epc = EP / WAVE >= 1 ? (EP + WAVE - 1) / WAVE : 1;
  for (int i = 0; i < 8; ++i) { acc_dg[i] = 0.f; acc_db[i] = 0.f; }

  long row = (long)blockIdx.x * waves_per_block + wave;
  const long stride = (long)gridDim.x * waves_per_block;
  for (; row < M; row += stride) {
    const float mu = mean[row];
    const float rs = rstd[row];
    const bf16* drow = dout + row * EP;
    const bf16* zrow = z + row * EP;
    bf16* dzrow = dz + row * EP;
    float s1 = 0.f, s2 = 0.f;
    // pass 1: h = du * gamma sums
    for (int i = 0; i < epc; ++i) {
      const int col = lane + i * WAVE;
      if (col >= E) continue;
      const float g = gamma[col];
      const float zv = bf2f(zrow[col]);
      const float xhat = (zv - mu) * rs;
      const float u = xhat * g + beta[col];
      const float y = tanhf(u);
      float dy = bf2f(drow[col]);
      if (p > 0.0f) {
        const float u01 =
            rng_uniform(seed, offset + (unsigned long long)row * EP + col);
        dy = (u01 >= p) ? dy * inv1mp : 0.0f;
      }
      const float du = dy * (1.0f - y * y);
      acc_dg[i] += du * xhat;
      acc_db[i] += du;
      const float h = du * g;
      s1 += h;
      s2 += h * xhat;
    }
    s1 = wave_reduce_sum(s1) * invE;
    s2 = wave_reduce_sum(s2) * invE;
    // pass 2: dz
    for (int i = 0; i < epc; ++i) {
      const int col = lane + i * WAVE;
      if (col >= EP) continue;
      float out_v = 0.f;
      if (col < E) {
        const float g = gamma[col];
        const float zv = bf2f(zrow[col]);
        const float xhat = (zv - mu) * rs;
        const float u = xhat * g + beta[col];
        const float y = tanhf(u);
        float dy = bf2f(drow[col]);
        if (p > 0.0f) {
          const float u01 =
              rng_uniform(seed, offset + (unsigned long long)row * EP + col);
          dy = (u01 >= p) ? dy * inv1mp : 0.0f;
        }
        const float du = dy * (1.0f - y * y);
        const float h = du * g;
        out_v = rs * (h - s1 - xhat * s2);
      }
      dzrow[col] = f2bf(out_v);
    }
  }
  // flush per-lane dgamma/dbeta partials (one atomic per lane per column)
  for (int i = 0; i < epc; ++i) {
    const int col = lane + i * WAVE;
    if (col < E) {
      atomic_add_f32(dgamma_part + col, acc_dg[i]);
      atomic_add_f32(dbeta_part + col, acc_db[i]);
    }
  }
}

extern "C" {

void launch_combiner_fwd(const void* X, const void* W, const float* gamma,
                         const float* beta, void* out, void* z, float* mean,
                         float* rstd, long M, int KP, int EP, int E, float p,
                         unsigned long long seed, unsigned long long offset,
                         hipStream_t stream) {
  const int NT = EP / 16;
  const long grid = (M + 127) / 128;
  const float inv1mp = p > 0.f ? 1.0f / (1.0f - p) : 1.0f;
  const int smem = 2 * NT * 64 * 8 * sizeof(bf16) + 2 * EP * sizeof(float);
#define CASE(nt)                                                              \
  case nt:                                                                    \
    combiner_fwd_kernel<nt><<<grid, 256, smem, stream>>>(                     \
        (const bf16*)X, (const bf16*)W, gamma, beta, (bf16*)out, (bf16*)z,    \
        mean, rstd, M, KP, E, p, inv1mp, seed, offset);                       \
    break;
  switch (NT) {
    CASE(2) CASE(4) CASE(6) CASE(8) CASE(10)
    CASE(12) CASE(14) CASE(16) CASE(18) CASE(20)
    default:
      printf("combiner_fwd: unsupported EP=%d\n", EP);
  }
#undef CASE
}

void launch_combiner_bwd(const void* dout, const void* z, const float* mean,
                         const float* rstd, const float* gamma,
                         const float* beta, void* dz, float* dgamma,
                         float* dbeta, long M, int EP, int E, float p,
                         unsigned long long seed, unsigned long long offset,
                         hipStream_t stream) {
  const int block = 256;
  const int wpb = block / WAVE;
  const int grid = (int)min((M + wpb - 1) / wpb, (long)8192);
  const float inv1mp = p > 0.f ? 1.0f / (1.0f - p) : 1.0f;
  combiner_bwd_kernel<<<grid, block, 0, stream>>>(
      (const bf16*)dout, (const bf16*)z, mean, rstd, gamma, beta, (bf16*)dz,
      dgamma, dbeta, M, EP, E, p, inv1mp, seed, offset);
}

}  // extern "C"
