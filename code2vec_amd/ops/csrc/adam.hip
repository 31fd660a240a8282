// K16: fused Adam step (semantics of torch.optim.Adam, reference main.py:138).
//   g = grad + wd * p;  m = b1*m + (1-b1)*g;  v = b2*v + (1-b2)*g*g
//   p -= lr * (m/bc1) / (sqrt(v/bc2) + eps)
// bf16 params keep an fp32 master (updated in fp32, rounded once); moments
// are fp32.  Vectorized 4-wide; one kernel per parameter tensor.

#include "common.h"

__global__ void adam_bf16_kernel(bf16* __restrict__ p,
                                 const bf16* __restrict__ g,
                                 float* __restrict__ master,
                                 float* __restrict__ m, float* __restrict__ v,
                                 long n, float lr, float b1, float b2,
                                 float eps, float wd, float inv_bc1,
                                 float inv_bc2) {
  const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 4;
  const long stride = (long)gridDim.x * blockDim.x * 4;
  for (long i = i0; i < n; i += stride) {
    if (i + 4 <= n) {
      float4 mm = *(float4*)(m + i);
      float4 vv = *(float4*)(v + i);
      float4 pp = *(float4*)(master + i);
      bf16 gg[4];
      *(ushort2*)gg = *(const ushort2*)(g + i);
      *(((ushort2*)gg) + 1) = *(((const ushort2*)(g + i)) + 1);
      bf16 pout[4];
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        float grad = bf2f(gg[k]);
        float* mk = k == 0 ? &mm.x : k == 1 ? &mm.y : k == 2 ? &mm.z : &mm.w;
        float* vk = k == 0 ? &vv.x : k == 1 ? &vv.y : k == 2 ? &vv.z : &vv.w;
        float* pk = k == 0 ? &pp.x : k == 1 ? &pp.y : k == 2 ? &pp.z : &pp.w;
        grad += wd * (*pk);
        *mk = b1 * (*mk) + (1.f - b1) * grad;
        *vk = b2 * (*vk) + (1.f - b2) * grad * grad;
        const float mhat = (*mk) * inv_bc1;
        const float vhat = (*vk) * inv_bc2;
        *pk -= lr * mhat / (sqrtf(vhat) + eps);
        pout[k] = f2bf(*pk);
      }
      *(float4*)(m + i) = mm;
      *(float4*)(v + i) = vv;
      *(float4*)(master + i) = pp;
      *(ushort2*)(p + i) = *(ushort2*)pout;
      *(((ushort2*)(p + i)) + 1) = *(((ushort2*)pout) + 1);
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
  const int grid = (int)min(want > 0 ? want : 1, (long)4096);
  adam_bf16_kernel<<<grid, block, 0, stream>>>((bf16*)p, (const bf16*)g,
                                               master, m, v, n, lr, b1, b2,
                                               eps, wd, inv_bc1, inv_bc2);
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
