// K11: ArcFace-style angular-margin head (reference model/model.py:71-80):
//   cos   = normalize(cv) @ normalize(W)^T            [B, L]
//   phi   = cos*cos_m - sin*sin_m,  phi where cos > 0 else cos
//   out   = (one_hot(y)*phi + (1-one_hot)*cos) * inverse_temp
// plus the building blocks of its backward:
//   dcos  = dout * s              (non-target columns)
//         = dout * s * dphi/dcos  (target column; clamp'd-sine gradient)
//   d(x/||x||) = (du - (du.u)u) / ||x||   (norm_project_kernel)
// The two backward GEMMs (du = dcos @ U_w, dv = dcos^T @ U_cv) reuse the
// existing head_dgrad / head_wgrad kernels (ops/functional.py).
//
// EP = 128 only (the model's padded encode width); other shapes take the
// torch fallback in ops/functional.py.

#include "common.h"

// inv[r] = 1 / max(||x[r, :]||, eps)  (F.normalize semantics, eps=1e-12)
__global__ __launch_bounds__(256) void inv_rownorm_kernel(
    const bf16* __restrict__ x, float* __restrict__ inv, long N) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const long r = (long)blockIdx.x * 4 + wave;
  if (r >= N) return;
  const bf16* row = x + r * 128;
  float ss = 0.f;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const float v = bf2f(row[lane * 2 + i]);
    ss += v * v;
  }
  ss = wave_reduce_sum(ss);
  if (lane == 0) inv[r] = 1.0f / fmaxf(sqrtf(ss), 1e-12f);
}

// out[r, :] = x[r, :] * inv[r]  (materialize the unit-row matrices)
__global__ __launch_bounds__(256) void rowscale_kernel(
    const bf16* __restrict__ x, const float* __restrict__ inv,
    bf16* __restrict__ out, long N) {
  const long i = (long)blockIdx.x * 256 + threadIdx.x;
  if (i >= N * 128) return;
  out[i] = f2bf(bf2f(x[i]) * inv[i / 128]);
}

// cos = U_cv @ U_w^T with the margin epilogue; emits BOTH the scaled
// outputs and the raw cosine (saved for backward).  Tile: 64 labels x
// 64 batch per block, 4 waves; both fragment loads are contiguous rows.
__global__ __launch_bounds__(256) void angular_fwd_kernel(
    const bf16* __restrict__ ucv, const bf16* __restrict__ uw,
    const long* __restrict__ label, bf16* __restrict__ out,
    bf16* __restrict__ cos_out, long B, long L, float cos_m, float sin_m,
    float s) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int GYB = (int)((B + 63) / 64);
  const long l0 = (long)(blockIdx.x / GYB) * 64;
  const long b0 = (long)(blockIdx.x % GYB) * 64;
  const int kj = (lane >> 4) * 8;

  const long arow = l0 + wave * 16 + (lane & 15);
  const bool aok = arow < L;
  const bf16* ap = uw + arow * 128 + kj;
  const bf16x8 zero8 = {};

  f32x4 acc[4];
#pragma unroll
  for (int nt = 0; nt < 4; ++nt) acc[nt] = f32x4{0.f, 0.f, 0.f, 0.f};

#pragma unroll
  for (int kk = 0; kk < 4; ++kk) {
    const bf16x8 a = aok ? *(const bf16x8*)(ap + kk * 32) : zero8;
#pragma unroll
    for (int nt = 0; nt < 4; ++nt) {
      const long bc = b0 + nt * 16 + (lane & 15);
      const bf16x8 b = (bc < B)
          ? *(const bf16x8*)(ucv + bc * 128 + kk * 32 + kj) : zero8;
      acc[nt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[nt],
                                                        0, 0, 0);
    }
  }

#pragma unroll
  for (int nt = 0; nt < 4; ++nt) {
    const long bc = b0 + nt * 16 + (lane & 15);
    if (bc >= B) continue;
    const long y = label[bc];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long lab = l0 + wave * 16 + (lane >> 4) * 4 + r;
      if (lab >= L) continue;
      const float c = acc[nt][r];
      float val = c;
      if (lab == y && c > 0.f) {
        const float sine = sqrtf(fmaxf(1.0f - c * c, 0.0f));
        val = c * cos_m - sine * sin_m;
      }
      out[bc * L + lab] = f2bf(val * s);
      cos_out[bc * L + lab] = f2bf(c);
    }
  }
}

// dcos from dout + saved cosine (elementwise over [B, L])
__global__ __launch_bounds__(256) void angular_dcos_kernel(
    const bf16* __restrict__ dout, const bf16* __restrict__ cosm,
    const long* __restrict__ label, bf16* __restrict__ dcos, long B, long L,
    float cos_m, float sin_m, float s) {
  const long i = (long)blockIdx.x * 256 + threadIdx.x;
  if (i >= B * L) return;
  const long b = i / L;
  const long l = i % L;
  float g = bf2f(dout[i]) * s;
  if (l == label[b]) {
    const float c = bf2f(cosm[i]);
    if (c > 0.f) {
      const float d2 = 1.0f - c * c;
      // clamp(1-c^2, min=0) passes gradient only where positive
      const float dsine = d2 > 0.f ? -c / sqrtf(fmaxf(d2, 1e-12f)) : 0.f;
      g *= cos_m + sin_m * (-dsine);
    }
  }
  dcos[i] = f2bf(g);
}

// out[r,:] = (du[r,:] - dot(du[r,:], u[r,:]) * u[r,:]) * inv[r]
__global__ __launch_bounds__(256) void norm_project_kernel(
    const bf16* __restrict__ du, const bf16* __restrict__ u,
    const float* __restrict__ inv, bf16* __restrict__ out, long N) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const long r = (long)blockIdx.x * 4 + wave;
  if (r >= N) return;
  const bf16* drow = du + r * 128;
  const bf16* urow = u + r * 128;
  float d0 = bf2f(drow[lane * 2]);
  float d1 = bf2f(drow[lane * 2 + 1]);
  const float u0 = bf2f(urow[lane * 2]);
  const float u1 = bf2f(urow[lane * 2 + 1]);
  const float dot = wave_reduce_sum(d0 * u0 + d1 * u1);
  const float iv = inv[r];
  out[r * 128 + lane * 2] = f2bf((d0 - dot * u0) * iv);
  out[r * 128 + lane * 2 + 1] = f2bf((d1 - dot * u1) * iv);
}

extern "C" {

void launch_inv_rownorm(const void* x, float* inv, long N,
                        hipStream_t stream) {
  inv_rownorm_kernel<<<(N + 3) / 4, 256, 0, stream>>>((const bf16*)x, inv,
                                                      N);
}

void launch_rowscale(const void* x, const float* inv, void* out, long N,
                     hipStream_t stream) {
  rowscale_kernel<<<(N * 128 + 255) / 256, 256, 0, stream>>>(
      (const bf16*)x, inv, (bf16*)out, N);
}

void launch_angular_fwd(const void* ucv, const void* uw, const long* label,
                        void* out, void* cos_out, long B, long L,
                        float cos_m, float sin_m, float s,
                        hipStream_t stream) {
  const long GXL = (L + 63) / 64;
  const long GYB = (B + 63) / 64;
  angular_fwd_kernel<<<GXL * GYB, 256, 0, stream>>>(
      (const bf16*)ucv, (const bf16*)uw, label, (bf16*)out, (bf16*)cos_out,
      B, L, cos_m, sin_m, s);
}

void launch_angular_dcos(const void* dout, const void* cosm,
                         const long* label, void* dcos, long B, long L,
                         float cos_m, float sin_m, float s,
                         hipStream_t stream) {
  angular_dcos_kernel<<<(B * L + 255) / 256, 256, 0, stream>>>(
      (const bf16*)dout, (const bf16*)cosm, label, (bf16*)dcos, B, L, cos_m,
      sin_m, s);
}

void launch_norm_project(const void* du, const void* u, const float* inv,
                         void* out, long N, hipStream_t stream) {
  norm_project_kernel<<<(N + 3) / 4, 256, 0, stream>>>(
      (const bf16*)du, (const bf16*)u, inv, (bf16*)out, N);
}

}  // extern "C"
