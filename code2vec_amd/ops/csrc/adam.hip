// K16: fused Adam step (semantics of torch.optim.Adam, reference main.py:138).
//   g = grad + wd * p;  m = b1*m + (1-b1)*g;  v = b2*v + (1-b2)*g*g
//   p -= lr * (m/bc1) / (sqrt(v/bc2) + eps)
// bf16 params keep an fp32 master (updated in fp32, rounded once); moments
// are fp32.  Vectorized 4-wide; one kernel per parameter tensor.

#include "common.h"

template <int NT>  // NT=1: nontemporal loads/stores for the streamed state
__global__ void adam_bf16_kernel(bf16* __restrict__ p,
                                 const bf16* __restrict__ g,
                                 float* __restrict__ master,
                                 float* __restrict__ m, float* __restrict__ v,
                                 long n, float lr, float b1, float b2,
                                 float eps, float wd, float inv_bc1,
                                 float inv_bc2) {
  const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  const long stride = (long)gridDim.x * blockDim.x * 4;
  typedef float f32x4v __attribute__((ext_vector_type(4)));
  typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
  for (long i = i0; i < n; i += stride) {
    if (i + 4 <= n) {
      f32x4v mm, vv, pp;
      bf16x4 gg;
      if (NT) {
        mm = __builtin_nontemporal_load((const f32x4v*)(m + i));
        vv = __builtin_nontemporal_load((const f32x4v*)(v + i));
        pp = __builtin_nontemporal_load((const f32x4v*)(master + i));
        gg = __builtin_nontemporal_load((const bf16x4*)(g + i));
      } else {
        mm = *(const f32x4v*)(m + i);
        vv = *(const f32x4v*)(v + i);
        pp = *(const f32x4v*)(master + i);
        gg = *(const bf16x4*)(g + i);
      }
      bf16x4 pout;
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        float grad = bf2f(gg[k]) + wd * pp[k];
        mm[k] = b1 * mm[k] + (1.f - b1) * grad;
        vv[k] = b2 * vv[k] + (1.f - b2) * grad * grad;
        pp[k] -= lr * (mm[k] * inv_bc1) / (sqrtf(vv[k] * inv_bc2) + eps);
        pout[k] = f2bf(pp[k]);
      }
      if (NT) {
        __builtin_nontemporal_store(mm, (f32x4v*)(m + i));
        __builtin_nontemporal_store(vv, (f32x4v*)(v + i));
        __builtin_nontemporal_store(pp, (f32x4v*)(master + i));
        __builtin_nontemporal_store(pout, (bf16x4*)(p + i));
      } else {
        *(f32x4v*)(m + i) = mm;
        *(f32x4v*)(v + i) = vv;
        *(f32x4v*)(master + i) = pp;
        *(bf16x4*)(p + i) = pout;
      }
    } else {
      for (long j = i; j < n; ++j) {
        float grad = bf2f(g[j]) + wd * master[j];
        m[j] = b1 * m[j] + (1.f - b1) * grad;
        v[j] = b2 * v[j] + (1.f - b2) * grad * grad;
        master[j] -= lr * (m[j] * inv_bc1) / (sqrtf(v[j] * inv_bc2) + eps);
        p[j] = f2bf(master[j]);
      }
    }
  }
}

__global__ void adam_f32_kernel(float* __restrict__ p,
                                const float* __restrict__ g,
                                float* __restrict__ m, float* __restrict__ v,
                                long n, float lr, float b1, float b2,
                                float eps, float wd, float inv_bc1,
                                float inv_bc2) {
  const long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long j = i; j < n; j += stride) {
    float grad = g[j] + wd * p[j];
    m[j] = b1 * m[j] + (1.f - b1) * grad;
    v[j] = b2 * v[j] + (1.f - b2) * grad * grad;
    p[j] -= lr * (m[j] * inv_bc1) / (sqrtf(v[j] * inv_bc2) + eps);
  }
}

extern "C" {

void launch_adam_bf16(void* p, const void* g, float* master, float* m,
                      float* v, long n, int step, float lr, float b1,
                      float b2, float eps, float wd, hipStream_t stream) {
  const float inv_bc1 = 1.0f / (1.0f - powf(b1, (float)step));
  const float inv_bc2 = 1.0f / (1.0f - powf(b2, (float)step));
  const int block = 256;
  const long want = (n / 4 + block - 1) / block;
  const int grid = (int)min(want > 0 ? want : 1, (long)8192);
  const char* nt_env = getenv("C2V_ADAM_NT");
  if (nt_env == nullptr || nt_env[0] != '0')
    adam_bf16_kernel<1><<<grid, block, 0, stream>>>(
        (bf16*)p, (const bf16*)g, master, m, v, n, lr, b1, b2, eps, wd,
        inv_bc1, inv_bc2);
  else
    adam_bf16_kernel<0><<<grid, block, 0, stream>>>(
        (bf16*)p, (const bf16*)g, master, m, v, n, lr, b1, b2, eps, wd,
        inv_bc1, inv_bc2);
}

void launch_adam_f32(float* p, const float* g, float* m, float* v, long n,
                     int step, float lr, float b1, float b2, float eps,
                     float wd, hipStream_t stream) {
  const float inv_bc1 = 1.0f / (1.0f - powf(b1, (float)step));
  const float inv_bc2 = 1.0f / (1.0f - powf(b2, (float)step));
  const int block = 256;
  const long want = (n + block - 1) / block;
  const int grid = (int)min(want > 0 ? want : 1, (long)4096);
  adam_f32_kernel<<<grid, block, 0, stream>>>(p, g, m, v, n, lr, b1, b2, eps,
                                              wd, inv_bc1, inv_bc2);
}

}  // extern "C"
