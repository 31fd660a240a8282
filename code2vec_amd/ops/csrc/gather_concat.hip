// K1/K2 fused triple embedding gather + concat, and K13 scatter-add backward.
// (reference op sites: model/model.py:48-51 forward; autograd of
// nn.Embedding for backward — reimagined as HBM-streaming CDNA4 kernels.)
//
// Layout: out[row, :] = [ term[starts[row]] | path[paths[row]] | term[ends[row]] ]
// with segment strides TS / PS / TS (each a multiple of 32 bf16 = 64 B), so
// every lane moves aligned 16-byte chunks.  Pad columns of the tables are
// zero, so pad columns of the output are zero by construction.

#include "common.h"

// One wave per output row (context); 4 waves per block; grid-stride over rows.
// Each lane copies 16-byte chunks; chunk -> segment resolved per lane.
__global__ void gather_concat_fwd_kernel(
    const int* __restrict__ starts, const int* __restrict__ paths,
    const int* __restrict__ ends, const bf16* __restrict__ term,
    const bf16* __restrict__ path, bf16* __restrict__ out,
    long M, int TS, int PS) {
  const int KP = 2 * TS + PS;
  const int chunks = KP / 8;  // 16-B chunks per row
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  long row = (long)blockIdx.x * waves_per_block + wave;
  const long stride = (long)gridDim.x * waves_per_block;

  for (; row < M; row += stride) {
    const long s = starts[row];
    const long p = paths[row];
    const long e = ends[row];
    const uint4* srow = (const uint4*)(term + s * TS);
    const uint4* prow = (const uint4*)(path + p * PS);
    const uint4* erow = (const uint4*)(term + e * TS);
    uint4* orow = (uint4*)(out + row * KP);
    const int ts8 = TS / 8, ps8 = PS / 8;
    for (int c = lane; c < chunks; c += WAVE) {
      uint4 v;
      if (c < ts8) v = srow[c];
      else if (c < ts8 + ps8) v = prow[c - ts8];
      else v = erow[c - ts8 - ps8];
      orow[c] = v;
    }
  }
}

// Backward: scatter-add grad rows into fp32 dense grad tables.
// Pad contexts (starts == 0) carry exactly-zero grads (attention mask math)
// and are skipped entirely.  fp32 accumulation via native atomics.
__global__ void gather_concat_bwd_kernel(
    const int* __restrict__ starts, const int* __restrict__ paths,
    const int* __restrict__ ends, const bf16* __restrict__ gout,
    float* __restrict__ dterm, float* __restrict__ dpath,
    long M, int TS, int PS) {
  const int KP = 2 * TS + PS;
  const int lane = threadIdx.x & (WAVE - 1);
  const int wave = threadIdx.x / WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  long row = (long)blockIdx.x * waves_per_block + wave;
  const long stride = (long)gridDim.x * waves_per_block;
  const int chunks = KP / 4;  // 4 bf16 (8 B) per lane-chunk

  for (; row < M; row += stride) {
    const int s = starts[row];
    if (s == 0) continue;  // pad context: grad identically zero
    const long p = paths[row];
    const long e = ends[row];
    const bf16* grow = gout + row * KP;
    const int ts4 = TS / 4, ps4 = PS / 4;
    for (int c = lane; c < chunks; c += WAVE) {
      float* dst;
      int col4;
      if (c < ts4) { dst = dterm + (long)s * TS; col4 = c; }
      else if (c < ts4 + ps4) { dst = dpath + p * PS; col4 = c - ts4; }
      else { dst = dterm + e * TS; col4 = c - ts4 - ps4; }
      // 4 bf16 grads -> 4 fp32 atomic adds
      const bf16* gsrc = grow + c * 4;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float g = bf2f(gsrc[j]);
        if (g != 0.0f) atomic_add_f32(dst + col4 * 4 + j, g);
      }
    }
  }
}

extern "C" {

void launch_gather_concat_fwd(const int* starts, const int* paths,
                              const int* ends, const void* term,
                              const void* path, void* out, long M, int TS,
                              int PS, hipStream_t stream) {
  const int block = 256;
  const int waves_per_block = block / WAVE;
  int grid = (int)min((M + waves_per_block - 1) / waves_per_block, (long)16384);
  gather_concat_fwd_kernel<<<grid, block, 0, stream>>>(
      starts, paths, ends, (const bf16*)term, (const bf16*)path, (bf16*)out,
      M, TS, PS);
}

void launch_gather_concat_bwd(const int* starts, const int* paths,
                              const int* ends, const void* gout, float* dterm,
                              float* dpath, long M, int TS, int PS,
                              hipStream_t stream) {
  const int block = 256;
  const int waves_per_block = block / WAVE;
  int grid = (int)min((M + waves_per_block - 1) / waves_per_block, (long)16384);
  gather_concat_bwd_kernel<<<grid, block, 0, stream>>>(
      starts, paths, ends, (const bf16*)gout, dterm, dpath, M, TS, PS);
}

}  // extern "C"
