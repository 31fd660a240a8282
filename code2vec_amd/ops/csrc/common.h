// Shared helpers for the code2vec_amd CDNA4 (gfx950) kernels.
// Wavefront = 64 lanes everywhere; block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE 64

typedef __bf16 bf16;
typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef __bf16 bf16x2 __attribute__((ext_vector_type(2)));
typedef __bf16 bf16x4 __attribute__((ext_vector_type(4)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

__device__ __forceinline__ float bf2f(bf16 x) { return (float)x; }
__device__ __forceinline__ bf16 f2bf(float x) { return (bf16)x; }

// ---------------------------------------------------------------------------
// Counter-based RNG for dropout masks: deterministic, stateless, cheap
// (32-bit finalizer; the backward never replays it — the mask is recovered
// from the saved forward output, so only mask-distribution quality matters).
__device__ __forceinline__ float rng_uniform(unsigned long long seed,
                                             unsigned long long idx) {
  unsigned int h = (unsigned int)(idx ^ (idx >> 32)) * 0x9E3779B9u;
  h ^= (unsigned int)seed ^ (unsigned int)(seed >> 32) * 0x85EBCA6Bu;
  h ^= h >> 16;
  h *= 0x7FEB352Du;
  h ^= h >> 15;
  h *= 0x846CA68Bu;
  h ^= h >> 16;
  return (float)(h >> 8) * (1.0f / 16777216.0f);
}

// tanh via one v_exp (exact identity; skips libm's precise-path branches):
// tanh(x) = sign(x) * (1 - 2/(e^{2|x|} + 1))
__device__ __forceinline__ float fast_tanh(float x) {
  const float ax = fabsf(x);
  const float t = 1.0f - 2.0f / (__expf(2.0f * ax) + 1.0f);
  return copysignf(t, x);
}

// ---------------------------------------------------------------------------
// Wave reductions (64-wide).
__device__ __forceinline__ float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off);
  return x;
}

__device__ __forceinline__ float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x = fmaxf(x, __shfl_xor(x, off));
  return x;
}

// Reduce within 16-lane groups (lanes l..l+15 with the same l>>4).
__device__ __forceinline__ float group16_reduce_sum(float x) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) x += __shfl_xor(x, off);
  return x;
}

// fp32 atomic add using the native global_atomic_add_f32 (no CAS loop).
__device__ __forceinline__ void atomic_add_f32(float* p, float v) {
  unsafeAtomicAdd(p, v);
}

#define C2V_CHECK_LAUNCH()                                                     \
  do {                                                                         \
    hipError_t e_ = hipGetLastError();                                         \
    if (e_ != hipSuccess) {                                                    \
      printf("kernel launch failed: %s\n", hipGetErrorString(e_));             \
    }                                                                          \
  } while (0)
