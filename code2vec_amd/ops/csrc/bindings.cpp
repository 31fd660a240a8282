// Python bindings for the code2vec_amd HIP kernel library (_c2v_hip).
// Pure C++ TU: declares the extern "C" launchers implemented in the .hip
// files and validates tensor dtype/contiguity/shape before launching on the
// current torch HIP stream.

#include <torch/extension.h>

#include <ATen/hip/HIPContext.h>
#include <hip/hip_runtime.h>

#define CHK(x) TORCH_CHECK(x, #x)
#define CHK_CONTIG(t) TORCH_CHECK((t).is_contiguous(), #t " must be contiguous")
#define CHK_CUDA(t) TORCH_CHECK((t).is_cuda(), #t " must be on GPU")
#define CHK_DT(t, dt) TORCH_CHECK((t).scalar_type() == dt, #t " dtype mismatch")

extern "C" {
void launch_gather_concat_fwd(const int*, const int*, const int*, const void*,
                              const void*, void*, long, int, int, int,
                              hipStream_t);
void launch_gather_concat_bwd(const int*, const int*, const int*, const void*,
                              float*, float*, long, int, int, hipStream_t);
void launch_embed_scatter_sorted(const int*, const long*, const void*, float*,
                                 void*, unsigned char*, long, long, int, int,
                                 int, int, int, hipStream_t);
void launch_count_indices(const int*, int*, long, hipStream_t);
void launch_cast_clear_rows(float*, const int*, unsigned char*, void*, long,
                            int, hipStream_t);
void launch_scatter_group(const int*, int*, int*, long*, long, hipStream_t);
void launch_exclusive_scan(const int*, int*, int*, long, hipStream_t);
size_t cub_exclusive_scan_temp_bytes(long);
void launch_cub_exclusive_scan(const int*, int*, void*, size_t, long,
                               hipStream_t);
void launch_combiner_fwd(const void*, const void*, const float*, const float*,
                         void*, void*, float*, float*, long, int, int, int,
                         float, unsigned long long,
                         const unsigned long long*, int, hipStream_t);
void launch_bump_u64(void*, unsigned long long, hipStream_t);
void launch_gather_combiner_fwd(const int*, const int*, const int*,
                                const void*, const void*, int, int,
                                const void*, const float*, const float*,
                                void*, void*, float*, float*, long, int, int,
                                int, float, unsigned long long,
                                const unsigned long long*, hipStream_t);
void launch_wgrad_gather(const int*, const int*, const int*, const void*,
                         const void*, int, int, const void*, float*, long,
                         int, int, int, hipStream_t);
void launch_combiner_bwd(const void*, const void*, const void*, const float*,
                         const float*, const float*, const float*, void*,
                         float*, float*, long, int, int, float, hipStream_t);
void launch_attention_fwd(const void*, const float*, const int*, float*,
                          float*, int, int, int, int, hipStream_t);
void launch_attention_bwd(const float*, const float*, const void*,
                          const float*, const int*, const float*, void*,
                          float*, int, int, int, int, int, hipStream_t);
void launch_lsm_nll_fwd(const void*, const long*, const float*, float*,
                        float*, int, long, hipStream_t);
void launch_lsm_nll_bwd(const void*, const long*, const float*, const float*,
                        const float*, const float*, void*, int, long,
                        hipStream_t);
void launch_wgrad(const void*, const void*, float*, long, int, int, int,
                  hipStream_t);
void launch_dgrad(const void*, const void*, void*, long, int, int,
                  hipStream_t);
void launch_head_wgrad(const void*, const void*, void*, long, long, int,
                       hipStream_t);
void launch_head_dgrad(const void*, const void*, float*, long, long,
                       hipStream_t);
void launch_transpose_w(const void*, void*, long, hipStream_t);
void launch_dgrad2(const void*, const void*, void*, long, long, hipStream_t);
void launch_slab_sum_bf16(const float*, void*, int, long, hipStream_t);
void launch_slab_sum_f32(const float*, float*, int, int, hipStream_t);
void launch_colsum_bf16(const void*, float*, long, long, hipStream_t);
void launch_row_max_argmax(const void*, float*, long*, long, long,
                           hipStream_t);
void launch_head_bwd_prep(const long*, const float*, const float*,
                          const float*, const float*, float*, long,
                          hipStream_t);
void launch_swizzle_cv(const void*, void*, long, long, hipStream_t);
void launch_inv_rownorm(const void*, float*, long, hipStream_t);
void launch_rowscale(const void*, const float*, void*, long, hipStream_t);
void launch_angular_fwd(const void*, const void*, const long*, void*, void*,
                        long, long, float, float, float, hipStream_t);
void launch_angular_dcos(const void*, const void*, const long*, void*, long,
                         long, float, float, float, hipStream_t);
void launch_norm_project(const void*, const void*, const float*, void*,
                         long, hipStream_t);
void launch_head_bwd_dw(const void*, const void*, const float*, void*,
                        float*, long, long, hipStream_t);
void launch_head_bwd_dcv_rc(const void*, const void*, const void*,
                            const float*, float*, long, long, int,
                            hipStream_t);
void launch_head_bwd_dcv(const void*, const void*, const float*, float*,
                         long, long, int, hipStream_t);
void launch_head_fwd(const void*, const void*, const float*, void*, float*,
                     float*, long, long, int, int, hipStream_t);
void launch_swizzle_a(const void*, void*, long, long, hipStream_t);
void launch_lsm_finalize(const void*, const float*, const float*, const long*,
                         const float*, float*, float*, int, long, int,
                         hipStream_t);
void launch_lsm_partial(const void*, float*, float*, int, long, hipStream_t);
void launch_adam_tick(double*, float, float, hipStream_t);
void launch_adam_f32_multi(float**, const float**, float**, float**,
                           const long*, int, const double*, float, float,
                           float, float, float, hipStream_t);
void launch_adam_bf16(void*, const void*, float*, float*, float*, long,
                      const double*, float, float, float, float, float,
                      hipStream_t);
void launch_adam_f32(float*, const float*, float*, float*, long,
                     const double*, float, float, float, float, float,
                     hipStream_t);
}

static hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void gather_concat_fwd(torch::Tensor starts, torch::Tensor paths,
                       torch::Tensor ends, torch::Tensor term,
                       torch::Tensor path, torch::Tensor out) {
  CHK_CUDA(starts); CHK_CONTIG(starts); CHK_DT(starts, torch::kInt32);
  CHK_CONTIG(paths); CHK_CONTIG(ends);
  CHK_DT(term, torch::kBFloat16); CHK_CONTIG(term);
  CHK_DT(path, torch::kBFloat16); CHK_CONTIG(path);
  CHK_DT(out, torch::kBFloat16); CHK_CONTIG(out);
  const long M = starts.numel();
  const int TS = term.size(1), PS = path.size(1);
  TORCH_CHECK(out.size(0) == M && out.size(1) >= 2 * TS + PS, "out shape");
  TORCH_CHECK(TS % 8 == 0 && PS % 8 == 0, "segment strides must be 8-mult");
  TORCH_CHECK(out.size(1) % 8 == 0, "out width must be 8-mult");
  launch_gather_concat_fwd(starts.data_ptr<int>(), paths.data_ptr<int>(),
                           ends.data_ptr<int>(), term.data_ptr(),
                           path.data_ptr(), out.data_ptr(), M, TS, PS,
                           (int)out.size(1), cur_stream());
}

void gather_concat_bwd(torch::Tensor starts, torch::Tensor paths,
                       torch::Tensor ends, torch::Tensor gout,
                       torch::Tensor dterm, torch::Tensor dpath) {
  CHK_CUDA(gout); CHK_CONTIG(gout); CHK_DT(gout, torch::kBFloat16);
  CHK_DT(dterm, torch::kFloat32); CHK_CONTIG(dterm);
  CHK_DT(dpath, torch::kFloat32); CHK_CONTIG(dpath);
  const long M = starts.numel();
  const int TS = dterm.size(1), PS = dpath.size(1);
  launch_gather_concat_bwd(starts.data_ptr<int>(), paths.data_ptr<int>(),
                           ends.data_ptr<int>(), gout.data_ptr(),
                           dterm.data_ptr<float>(), dpath.data_ptr<float>(),
                           M, TS, PS, cur_stream());
}

void embed_scatter_sorted(torch::Tensor sorted_idx, torch::Tensor perm,
                          torch::Tensor gout, torch::Tensor dtable,
                          torch::Tensor out_bf16, torch::Tensor flags,
                          int64_t M, int64_t KP, int64_t off0, int64_t off1,
                          int64_t R) {
  CHK_CUDA(sorted_idx); CHK_CONTIG(sorted_idx);
  CHK_DT(sorted_idx, torch::kInt32); CHK_DT(perm, torch::kInt64);
  CHK_DT(gout, torch::kBFloat16); CHK_CONTIG(gout);
  CHK_DT(dtable, torch::kFloat32); CHK_CONTIG(dtable);
  CHK_DT(out_bf16, torch::kBFloat16); CHK_CONTIG(out_bf16);
  CHK_DT(flags, torch::kUInt8);
  const long N = sorted_idx.numel();
  const int S = dtable.size(1);
  launch_embed_scatter_sorted(sorted_idx.data_ptr<int>(),
                              perm.data_ptr<long>(), gout.data_ptr(),
                              dtable.data_ptr<float>(), out_bf16.data_ptr(),
                              flags.data_ptr<unsigned char>(), N, M, (int)KP,
                              S, (int)off0, (int)off1, (int)R, cur_stream());
}

void cast_clear_rows(torch::Tensor dtable, torch::Tensor counts,
                     torch::Tensor flags, torch::Tensor out) {
  CHK_CUDA(dtable); CHK_CONTIG(dtable); CHK_DT(dtable, torch::kFloat32);
  CHK_DT(counts, torch::kInt32); CHK_DT(out, torch::kBFloat16);
  CHK_DT(flags, torch::kUInt8); CHK_CONTIG(out);
  const long T = dtable.size(0);
  const int S = dtable.size(1);
  TORCH_CHECK(counts.numel() >= T && flags.numel() >= T, "counts/flags");
  launch_cast_clear_rows(dtable.data_ptr<float>(), counts.data_ptr<int>(),
                         flags.data_ptr<unsigned char>(), out.data_ptr(), T,
                         S, cur_stream());
}

// rng_off: int64 [1] DEVICE scalar holding the dropout counter offset;
// consumed by the kernel and advanced by M*EP on the same stream (device-
// side state so hipGraph replays keep drawing fresh masks).
void combiner_fwd(torch::Tensor x, torch::Tensor w, torch::Tensor gamma,
                  torch::Tensor beta, torch::Tensor out, torch::Tensor z,
                  torch::Tensor mean, torch::Tensor rstd, int64_t E,
                  double p, int64_t seed, torch::Tensor rng_off,
                  int64_t epilogue_mode) {
  CHK_CUDA(x); CHK_CONTIG(x); CHK_DT(x, torch::kBFloat16);
  CHK_CONTIG(w); CHK_DT(w, torch::kBFloat16);
  CHK_DT(gamma, torch::kFloat32); CHK_DT(beta, torch::kFloat32);
  CHK_DT(out, torch::kBFloat16); CHK_DT(z, torch::kBFloat16);
  CHK_DT(mean, torch::kFloat32); CHK_DT(rstd, torch::kFloat32);
  const long M = x.size(0);
  const int KP = x.size(1), EP = w.size(0);  // w is TRANSPOSED: [EP, KP]
  TORCH_CHECK(w.size(1) == KP && KP % 32 == 0 && EP % 32 == 0, "w shape");
  CHK_DT(rng_off, torch::kInt64);
  launch_combiner_fwd(x.data_ptr(), w.data_ptr(), gamma.data_ptr<float>(),
                      beta.data_ptr<float>(), out.data_ptr(), z.data_ptr(),
                      mean.data_ptr<float>(), rstd.data_ptr<float>(), M, KP,
                      EP, (int)E, (float)p, (unsigned long long)seed,
                      (const unsigned long long*)rng_off.data_ptr(),
                      (int)epilogue_mode, cur_stream());
  if (p > 0.0)
    launch_bump_u64(rng_off.data_ptr(), (unsigned long long)(M * EP),
                    cur_stream());
}

void gather_combiner_fwd(torch::Tensor starts, torch::Tensor paths,
                         torch::Tensor ends, torch::Tensor term,
                         torch::Tensor path, torch::Tensor w,
                         torch::Tensor gamma, torch::Tensor beta,
                         torch::Tensor out, torch::Tensor z,
                         torch::Tensor mean, torch::Tensor rstd, int64_t KP,
                         int64_t E, double p, int64_t seed,
                         torch::Tensor rng_off) {
  CHK_CUDA(starts); CHK_CONTIG(starts); CHK_DT(starts, torch::kInt32);
  CHK_DT(term, torch::kBFloat16); CHK_CONTIG(term);
  CHK_DT(path, torch::kBFloat16); CHK_CONTIG(path);
  CHK_CONTIG(w); CHK_DT(w, torch::kBFloat16);
  CHK_DT(out, torch::kBFloat16); CHK_DT(z, torch::kBFloat16);
  const long M = starts.numel();
  const int EP = w.size(0);
  const int TS = term.size(1), PS = path.size(1);
  TORCH_CHECK(w.size(1) == KP && KP >= 2 * TS + PS, "shapes");
  CHK_DT(rng_off, torch::kInt64);
  launch_gather_combiner_fwd(
      starts.data_ptr<int>(), paths.data_ptr<int>(), ends.data_ptr<int>(),
      term.data_ptr(), path.data_ptr(), TS, PS, w.data_ptr(),
      gamma.data_ptr<float>(), beta.data_ptr<float>(), out.data_ptr(),
      z.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(), M,
      (int)KP, EP, (int)E, (float)p, (unsigned long long)seed,
      (const unsigned long long*)rng_off.data_ptr(), cur_stream());
  if (p > 0.0)
    launch_bump_u64(rng_off.data_ptr(),
                    (unsigned long long)(M * (long)w.size(0)),
                    cur_stream());
}

void wgrad_gather(torch::Tensor starts, torch::Tensor paths,
                  torch::Tensor ends, torch::Tensor term, torch::Tensor path,
                  torch::Tensor dz, torch::Tensor partials, int64_t KP) {
  CHK_CUDA(dz); CHK_CONTIG(dz); CHK_DT(dz, torch::kBFloat16);
  CHK_DT(partials, torch::kFloat32); CHK_CONTIG(partials);
  const long M = starts.numel();
  const int EP = dz.size(1);
  const int TS = term.size(1), PS = path.size(1);
  TORCH_CHECK(partials.size(1) == KP && partials.size(2) == EP, "partials");
  launch_wgrad_gather(starts.data_ptr<int>(), paths.data_ptr<int>(),
                      ends.data_ptr<int>(), term.data_ptr(), path.data_ptr(),
                      TS, PS, dz.data_ptr(), partials.data_ptr<float>(), M,
                      (int)KP, EP, partials.size(0), cur_stream());
}

void combiner_bwd(torch::Tensor dout, torch::Tensor z, torch::Tensor out,
                  torch::Tensor mean, torch::Tensor rstd, torch::Tensor gamma,
                  torch::Tensor beta, torch::Tensor dz, torch::Tensor dgamma,
                  torch::Tensor dbeta, int64_t E, double p) {
  CHK_CUDA(dout); CHK_CONTIG(dout); CHK_DT(dout, torch::kBFloat16);
  CHK_CONTIG(z); CHK_CONTIG(out); CHK_DT(out, torch::kBFloat16);
  CHK_DT(dz, torch::kBFloat16);
  const long M = z.size(0);
  const int EP = z.size(1);
  launch_combiner_bwd(dout.data_ptr(), z.data_ptr(), out.data_ptr(),
                      mean.data_ptr<float>(), rstd.data_ptr<float>(),
                      gamma.data_ptr<float>(), beta.data_ptr<float>(),
                      dz.data_ptr(), dgamma.data_ptr<float>(),
                      dbeta.data_ptr<float>(), M, EP, (int)E, (float)p,
                      cur_stream());
}

void group_by_index(torch::Tensor idx, torch::Tensor counts,
                    torch::Tensor cursor, torch::Tensor sorted_idx,
                    torch::Tensor perm, bool count_only) {
  CHK_CUDA(idx); CHK_CONTIG(idx); CHK_DT(idx, torch::kInt32);
  const long N = idx.numel();
  if (count_only) {
    launch_count_indices(idx.data_ptr<int>(), counts.data_ptr<int>(), N,
                         cur_stream());
  } else {
    launch_scatter_group(idx.data_ptr<int>(), cursor.data_ptr<int>(),
                         sorted_idx.data_ptr<int>(), perm.data_ptr<long>(), N,
                         cur_stream());
  }
}

void attention_fwd(torch::Tensor ccv, torch::Tensor a, torch::Tensor starts,
                   torch::Tensor cv, torch::Tensor attn, int64_t E) {
  CHK_CUDA(ccv); CHK_CONTIG(ccv); CHK_DT(ccv, torch::kBFloat16);
  CHK_DT(a, torch::kFloat32); CHK_DT(cv, torch::kFloat32);
  CHK_DT(attn, torch::kFloat32); CHK_DT(starts, torch::kInt32);
  const int B = ccv.size(0), C = ccv.size(1), EP = ccv.size(2);
  launch_attention_fwd(ccv.data_ptr(), a.data_ptr<float>(),
                       starts.data_ptr<int>(), cv.data_ptr<float>(),
                       attn.data_ptr<float>(), B, C, EP, (int)E,
                       cur_stream());
}

void attention_bwd(torch::Tensor dcv, torch::Tensor dattn, torch::Tensor ccv,
                   torch::Tensor a, torch::Tensor starts, torch::Tensor attn,
                   torch::Tensor dccv, torch::Tensor da, int64_t E,
                   bool has_dattn) {
  CHK_CUDA(dcv); CHK_CONTIG(dcv); CHK_DT(dcv, torch::kFloat32);
  CHK_DT(dccv, torch::kBFloat16); CHK_DT(da, torch::kFloat32);
  const int B = ccv.size(0), C = ccv.size(1), EP = ccv.size(2);
  launch_attention_bwd(dcv.data_ptr<float>(),
                       has_dattn ? dattn.data_ptr<float>() : nullptr,
                       ccv.data_ptr(), a.data_ptr<float>(),
                       starts.data_ptr<int>(), attn.data_ptr<float>(),
                       dccv.data_ptr(), da.data_ptr<float>(), B, C, EP,
                       (int)E, has_dattn ? 1 : 0, cur_stream());
}

void logsoftmax_nll_fwd(torch::Tensor logits, torch::Tensor label,
                        torch::Tensor weight, torch::Tensor lse,
                        torch::Tensor acc) {
  CHK_CUDA(logits); CHK_CONTIG(logits); CHK_DT(logits, torch::kBFloat16);
  CHK_DT(label, torch::kInt64); CHK_DT(lse, torch::kFloat32);
  const int B = logits.size(0);
  const long L = logits.size(1);
  TORCH_CHECK(acc.numel() == 2L * B, "acc must be [B, 2] partials");
  const float* wp = weight.defined() && weight.numel() > 0
                        ? weight.data_ptr<float>()
                        : nullptr;
  launch_lsm_nll_fwd(logits.data_ptr(), label.data_ptr<long>(), wp,
                     lse.data_ptr<float>(), acc.data_ptr<float>(), B, L,
                     cur_stream());
}

void logsoftmax_nll_bwd(torch::Tensor logits, torch::Tensor label,
                        torch::Tensor weight, torch::Tensor lse,
                        torch::Tensor acc, torch::Tensor gscale,
                        torch::Tensor dlogits) {
  CHK_CUDA(logits); CHK_CONTIG(dlogits); CHK_DT(dlogits, torch::kBFloat16);
  const int B = logits.size(0);
  const long L = logits.size(1);
  const float* wp = weight.defined() && weight.numel() > 0
                        ? weight.data_ptr<float>()
                        : nullptr;
  launch_lsm_nll_bwd(logits.data_ptr(), label.data_ptr<long>(), wp,
                     lse.data_ptr<float>(), acc.data_ptr<float>(),
                     gscale.data_ptr<float>(), dlogits.data_ptr(), B, L,
                     cur_stream());
}

// (max, sumexp) partials over 16384-col chunks -> pm/ps[GX][B]
void lsm_partial(torch::Tensor logits, torch::Tensor pm, torch::Tensor ps) {
  CHK_CUDA(logits); CHK_CONTIG(logits); CHK_DT(logits, torch::kBFloat16);
  CHK_DT(pm, torch::kFloat32); CHK_CONTIG(pm); CHK_CONTIG(ps);
  const int B = logits.size(0);
  const long L = logits.size(1);
  const long GX = (L + 16383) / 16384;
  TORCH_CHECK(pm.numel() == GX * B && ps.numel() == GX * B,
              "lsm_partial partials shape");
  launch_lsm_partial(logits.data_ptr(), pm.data_ptr<float>(),
                     ps.data_ptr<float>(), B, L, cur_stream());
}

void dgrad(torch::Tensor dz, torch::Tensor w2, torch::Tensor dx) {
  CHK_CUDA(dz); CHK_CONTIG(dz); CHK_DT(dz, torch::kBFloat16);
  CHK_CONTIG(w2); CHK_DT(w2, torch::kBFloat16);
  CHK_CONTIG(dx); CHK_DT(dx, torch::kBFloat16);
  const long M = dz.size(0);
  const int EP = dz.size(1), KP = w2.size(0);
  TORCH_CHECK(w2.size(1) == EP && dx.size(1) == KP, "dgrad shapes");
  launch_dgrad(dz.data_ptr(), w2.data_ptr(), dx.data_ptr(), M, KP, EP,
               cur_stream());
}

void head_wgrad(torch::Tensor dlogits, torch::Tensor cv, torch::Tensor dw) {
  CHK_CUDA(dlogits); CHK_CONTIG(dlogits); CHK_DT(dlogits, torch::kBFloat16);
  CHK_CONTIG(cv); CHK_DT(cv, torch::kBFloat16);
  CHK_CONTIG(dw); CHK_DT(dw, torch::kBFloat16);
  const long B = dlogits.size(0), L = dlogits.size(1);
  const int EP = cv.size(1);
  TORCH_CHECK(dw.size(0) == L && dw.size(1) == EP, "dw shape");
  TORCH_CHECK(B % 32 == 0 && L % 8 == 0, "head_wgrad shape gates");
  launch_head_wgrad(dlogits.data_ptr(), cv.data_ptr(), dw.data_ptr(), L, B,
                    EP, cur_stream());
}

// logits = cv @ w^T + bias; pm/ps (optional, pass undefined tensors to
// skip) receive per-(row, 64-col-block) online-softmax partials.
void head_fwd(torch::Tensor cv, torch::Tensor w, torch::Tensor bias,
              torch::Tensor out, torch::Tensor pm, torch::Tensor ps) {
  CHK_CUDA(cv); CHK_CONTIG(cv); CHK_DT(cv, torch::kBFloat16);
  CHK_CONTIG(w); CHK_DT(w, torch::kBFloat16);
  CHK_DT(bias, torch::kFloat32); CHK_CONTIG(bias);
  CHK_CONTIG(out); CHK_DT(out, torch::kBFloat16);
  const long B = cv.size(0), L = w.size(0);
  const int EP = cv.size(1);
  TORCH_CHECK(w.size(1) == EP && EP % 32 == 0, "head_fwd EP");
  TORCH_CHECK(out.size(0) == B && out.size(1) == L && bias.numel() == L,
              "head_fwd shapes");
  float* pmp = nullptr;
  float* psp = nullptr;
  if (pm.defined() && pm.numel() > 0) {
    const long GXL = (L + 255) / 256;  // [lab_block, B] layout
    CHK_DT(pm, torch::kFloat32); CHK_CONTIG(pm);
    TORCH_CHECK(pm.numel() == B * GXL && ps.numel() == B * GXL, "partials");
    pmp = pm.data_ptr<float>();
    psp = ps.data_ptr<float>();
  }
  launch_head_fwd(cv.data_ptr(), w.data_ptr(), bias.data_ptr<float>(),
                  out.data_ptr(), pmp, psp, B, L, EP, /*aimg=*/0,
                  cur_stream());
}

// head_fwd consuming the swizzle_a W image ([ceil(L/256)*16, 4, 64, 8]);
// out/pm/ps contracts as head_fwd
void head_fwd_img(torch::Tensor cv, torch::Tensor wimg, torch::Tensor bias,
                  torch::Tensor out, torch::Tensor pm, torch::Tensor ps,
                  int64_t L) {
  CHK_CUDA(cv); CHK_CONTIG(cv); CHK_DT(cv, torch::kBFloat16);
  CHK_CONTIG(wimg); CHK_DT(wimg, torch::kBFloat16);
  CHK_DT(bias, torch::kFloat32); CHK_CONTIG(bias);
  CHK_CONTIG(out); CHK_DT(out, torch::kBFloat16);
  const long B = cv.size(0);
  const int EP = cv.size(1);
  TORCH_CHECK(EP == 128, "head_fwd_img is EP=128 only");
  TORCH_CHECK(wimg.numel() == (L + 255) / 256 * 16 * 2048,
              "wimg must be [ceil(L/256)*16, 4, 64, 8]");
  TORCH_CHECK(out.size(0) == B && out.size(1) == L && bias.numel() == L,
              "head_fwd_img shapes");
  float* pmp = nullptr;
  float* psp = nullptr;
  if (pm.defined() && pm.numel() > 0) {
    const long GXL = (L + 255) / 256;
    CHK_DT(pm, torch::kFloat32); CHK_CONTIG(pm);
    TORCH_CHECK(pm.numel() == B * GXL && ps.numel() == B * GXL, "partials");
    pmp = pm.data_ptr<float>();
    psp = ps.data_ptr<float>();
  }
  launch_head_fwd(cv.data_ptr(), wimg.data_ptr(), bias.data_ptr<float>(),
                  out.data_ptr(), pmp, psp, B, L, EP, /*aimg=*/1,
                  cur_stream());
}

void swizzle_a(torch::Tensor w, torch::Tensor wimg) {
  CHK_CUDA(w); CHK_CONTIG(w); CHK_DT(w, torch::kBFloat16);
  CHK_CONTIG(wimg); CHK_DT(wimg, torch::kBFloat16);
  const long L = w.size(0);
  TORCH_CHECK(w.size(1) == 128, "w must be [L, 128]");
  const long nrt = wimg.numel() / 2048;
  TORCH_CHECK(nrt * 2048 == wimg.numel() && nrt >= (L + 15) / 16,
              "wimg must be [nrt >= ceil(L/16), 4, 64, 8]");
  launch_swizzle_a(w.data_ptr(), wimg.data_ptr(), L, nrt, cur_stream());
}

void logsoftmax_nll_finalize(torch::Tensor logits, torch::Tensor pm,
                             torch::Tensor ps, torch::Tensor label,
                             torch::Tensor weight, torch::Tensor lse,
                             torch::Tensor acc) {
  CHK_CUDA(logits); CHK_CONTIG(logits); CHK_DT(logits, torch::kBFloat16);
  CHK_DT(pm, torch::kFloat32); CHK_CONTIG(pm);
  CHK_DT(lse, torch::kFloat32);
  const int B = logits.size(0);
  const long L = logits.size(1);
  // GX derived from the partials shape: 256-col blocks from the head_fwd
  // epilogue, 4096-col blocks from lsm_partial — finalize is generic
  const int GX = (int)(pm.numel() / B);
  TORCH_CHECK((long)GX * B == pm.numel() && ps.numel() == pm.numel(),
              "partials");
  TORCH_CHECK(acc.numel() == 2L * ((B + 15) / 16),
              "acc must be [(B+15)/16, 2] partials");
  const float* wp = weight.defined() && weight.numel() > 0
                        ? weight.data_ptr<float>()
                        : nullptr;
  launch_lsm_finalize(logits.data_ptr(), pm.data_ptr<float>(),
                      ps.data_ptr<float>(), label.data_ptr<long>(), wp,
                      lse.data_ptr<float>(), acc.data_ptr<float>(), B, L, GX,
                      cur_stream());
}

// dcv split-K partials: partials[ceil(L/512), B, 128] f32; wt = W
// transposed [EP=128, L]
void head_dgrad(torch::Tensor dlogits, torch::Tensor wt,
                torch::Tensor partials) {
  CHK_CUDA(dlogits); CHK_CONTIG(dlogits); CHK_DT(dlogits, torch::kBFloat16);
  CHK_CONTIG(wt); CHK_DT(wt, torch::kBFloat16);
  CHK_DT(partials, torch::kFloat32); CHK_CONTIG(partials);
  const long B = dlogits.size(0), L = dlogits.size(1);
  TORCH_CHECK(wt.size(0) == 128 && wt.size(1) == L, "wt must be [128, L]");
  TORCH_CHECK(L % 8 == 0, "head_dgrad needs L % 8 == 0");
  TORCH_CHECK(partials.numel() == (L + 511) / 512 * B * 128, "partials");
  launch_head_dgrad(dlogits.data_ptr(), wt.data_ptr(),
                    partials.data_ptr<float>(), B, L, cur_stream());
}

// Fused head+loss backward prep: coef_lse[B, 4] = (coef_b, lse_b, y_b, 0).
// exclusive prefix scan x[0..n-1] -> out[0..n-1] (int32); partial is a
// scratch of >= ceil(n/1024) ints
void exclusive_scan(torch::Tensor x, torch::Tensor partial,
                    torch::Tensor out) {
  CHK_CUDA(x); CHK_CONTIG(x); CHK_DT(x, torch::kInt32);
  CHK_CONTIG(partial); CHK_DT(partial, torch::kInt32);
  CHK_CONTIG(out); CHK_DT(out, torch::kInt32);
  const long n = x.numel();
  TORCH_CHECK(out.numel() >= n, "scan out too small");
  TORCH_CHECK(partial.numel() >= (n + 1023) / 1024, "scan partial too small");
  launch_exclusive_scan(x.data_ptr<int>(), partial.data_ptr<int>(),
                        out.data_ptr<int>(), n, cur_stream());
}

long cub_scan_temp_bytes(long n) {
  return (long)cub_exclusive_scan_temp_bytes(n);
}

// hipCUB decoupled-lookback exclusive scan; temp is a uint8 scratch of
// at least cub_scan_temp_bytes(n)
void cub_exclusive_scan(torch::Tensor x, torch::Tensor temp,
                        torch::Tensor out) {
  CHK_CUDA(x); CHK_CONTIG(x); CHK_DT(x, torch::kInt32);
  CHK_CONTIG(out); CHK_DT(out, torch::kInt32);
  const long n = x.numel();
  TORCH_CHECK(out.numel() >= n, "scan out too small");
  TORCH_CHECK(temp.numel() >= (long)cub_exclusive_scan_temp_bytes(n),
              "cub scan temp too small");
  launch_cub_exclusive_scan(x.data_ptr<int>(), out.data_ptr<int>(),
                            temp.data_ptr(), (size_t)temp.numel(), n,
                            cur_stream());
}

void head_bwd_prep(torch::Tensor label, torch::Tensor weight,
                   torch::Tensor acc, torch::Tensor gscale, torch::Tensor lse,
                   torch::Tensor coef_lse) {
  CHK_CUDA(lse); CHK_DT(lse, torch::kFloat32); CHK_DT(label, torch::kInt64);
  CHK_CONTIG(coef_lse); CHK_DT(coef_lse, torch::kFloat32);
  const long B = label.numel();
  TORCH_CHECK(coef_lse.numel() == 4 * B, "coef_lse must be [B, 4]");
  const float* wp = weight.defined() && weight.numel() > 0
                        ? weight.data_ptr<float>()
                        : nullptr;
  launch_head_bwd_prep(label.data_ptr<long>(), wp, acc.data_ptr<float>(),
                       gscale.data_ptr<float>(), lse.data_ptr<float>(),
                       coef_lse.data_ptr<float>(), B, cur_stream());
}

// [N, 128] -> B-fragment image [nchunk, 8, 64, 8]; nchunk (from the
// output shape) must cover N at the consumer's stage granularity.
void swizzle_cv(torch::Tensor cv, torch::Tensor cvimg) {
  CHK_CUDA(cv); CHK_CONTIG(cv); CHK_DT(cv, torch::kBFloat16);
  CHK_CONTIG(cvimg); CHK_DT(cvimg, torch::kBFloat16);
  const long B = cv.size(0);
  TORCH_CHECK(cv.size(1) == 128, "operand must be [N, 128]");
  const long nchunk = cvimg.numel() / 4096;
  TORCH_CHECK(nchunk * 4096 == cvimg.numel() && nchunk >= (B + 31) / 32,
              "image shape must be [nchunk >= ceil(N/32), 8, 64, 8]");
  launch_swizzle_cv(cv.data_ptr(), cvimg.data_ptr(), B, nchunk,
                    cur_stream());
}

// Fused head+loss backward kernel 1: dw[L, 128] bf16 + dbias[L] f32 from
// (logits, coef_lse) with G recomputed in-kernel; cvimg = swizzle_cv(cv).
void head_bwd_dw(torch::Tensor logits, torch::Tensor cvimg,
                 torch::Tensor coef_lse, torch::Tensor dw,
                 torch::Tensor dbias) {
  CHK_CUDA(logits); CHK_CONTIG(logits); CHK_DT(logits, torch::kBFloat16);
  CHK_CONTIG(cvimg); CHK_DT(cvimg, torch::kBFloat16);
  CHK_CONTIG(coef_lse); CHK_DT(coef_lse, torch::kFloat32);
  CHK_CONTIG(dw); CHK_DT(dw, torch::kBFloat16);
  CHK_CONTIG(dbias); CHK_DT(dbias, torch::kFloat32);
  const long B = logits.size(0), L = logits.size(1);
  TORCH_CHECK(cvimg.numel() == (B + 63) / 64 * 2 * 4096, "cvimg shape");
  TORCH_CHECK(L % 8 == 0 && B % 8 == 0, "head_bwd_dw shape gates");
  TORCH_CHECK(L < (1L << 24), "label index carried as f32 needs L < 2^24");
  TORCH_CHECK(coef_lse.numel() == 4 * B, "coef_lse must be [B, 4]");
  TORCH_CHECK(dw.size(0) == L && dw.size(1) == 128 && dbias.numel() == L,
              "head_bwd_dw output shapes");
  launch_head_bwd_dw(logits.data_ptr(), cvimg.data_ptr(),
                     coef_lse.data_ptr<float>(), dw.data_ptr(),
                     dbias.data_ptr<float>(), B, L, cur_stream());
}

// Fused head+loss backward kernel 2: dcv split-K partials
// [ceil(L/chunk), B, 128] f32 with G recomputed; wimg = swizzle_cv(W)
// at 128-label granularity ([ceil(L/128)*4, 8, 64, 8]).
void head_bwd_dcv(torch::Tensor logits, torch::Tensor wimg,
                  torch::Tensor coef_lse, torch::Tensor partials,
                  long chunk) {
  CHK_CUDA(logits); CHK_CONTIG(logits); CHK_DT(logits, torch::kBFloat16);
  CHK_CONTIG(wimg); CHK_DT(wimg, torch::kBFloat16);
  CHK_CONTIG(coef_lse); CHK_DT(coef_lse, torch::kFloat32);
  CHK_DT(partials, torch::kFloat32); CHK_CONTIG(partials);
  const long B = logits.size(0), L = logits.size(1);
  TORCH_CHECK(wimg.numel() == (L + 127) / 128 * 4 * 4096,
              "wimg must be [ceil(L/128)*4, 8, 64, 8]");
  TORCH_CHECK(L % 8 == 0 && chunk % 128 == 0, "head_bwd_dcv shape gates");
  TORCH_CHECK(L < (1L << 24), "label index carried as f32 needs L < 2^24");
  TORCH_CHECK(coef_lse.numel() == 4 * B, "coef_lse must be [B, 4]");
  TORCH_CHECK(partials.numel() == (L + chunk - 1) / chunk * B * 128,
              "head_bwd_dcv partials shape");
  launch_head_bwd_dcv(logits.data_ptr(), wimg.data_ptr(),
                      coef_lse.data_ptr<float>(),
                      partials.data_ptr<float>(), B, L, (int)chunk,
                      cur_stream());
}

// Streaming dcv: logits never read — recomputed per wave from the
// swizzle_a fragment images of cv ([B,128]) and W ([L,128]).
void head_bwd_dcv_rc(torch::Tensor cvimg_a, torch::Tensor wimg_a,
                     torch::Tensor wimg, torch::Tensor coef_lse,
                     torch::Tensor partials, long B, long L, long chunk) {
  CHK_CUDA(wimg); CHK_CONTIG(wimg); CHK_DT(wimg, torch::kBFloat16);
  CHK_CONTIG(cvimg_a); CHK_DT(cvimg_a, torch::kBFloat16);
  CHK_CONTIG(wimg_a); CHK_DT(wimg_a, torch::kBFloat16);
  CHK_CONTIG(coef_lse); CHK_DT(coef_lse, torch::kFloat32);
  CHK_DT(partials, torch::kFloat32); CHK_CONTIG(partials);
  TORCH_CHECK(wimg.numel() == (L + 127) / 128 * 4 * 4096,
              "wimg must be [ceil(L/128)*4, 8, 64, 8]");
  TORCH_CHECK(cvimg_a.numel() >= (B + 127) / 128 * 8 * 2048,
              "cvimg_a too small for B");
  TORCH_CHECK(wimg_a.numel() >= (L + 255) / 256 * 16 * 2048,
              "wimg_a too small for L");
  TORCH_CHECK(L % 8 == 0 && chunk % 128 == 0, "head_bwd_dcv_rc shape gates");
  TORCH_CHECK(L < (1L << 24), "label index carried as f32 needs L < 2^24");
  TORCH_CHECK(coef_lse.numel() == 4 * B, "coef_lse must be [B, 4]");
  TORCH_CHECK(partials.numel() == (L + chunk - 1) / chunk * B * 128,
              "head_bwd_dcv_rc partials shape");
  launch_head_bwd_dcv_rc(cvimg_a.data_ptr(), wimg_a.data_ptr(),
                         wimg.data_ptr(), coef_lse.data_ptr<float>(),
                         partials.data_ptr<float>(), B, L, (int)chunk,
                         cur_stream());
}

// K11 angular-margin building blocks (all [N, 128] bf16 row spaces)
void inv_rownorm(torch::Tensor x, torch::Tensor inv) {
  CHK_CUDA(x); CHK_CONTIG(x); CHK_DT(x, torch::kBFloat16);
  CHK_DT(inv, torch::kFloat32);
  TORCH_CHECK(x.size(1) == 128 && inv.numel() == x.size(0), "inv_rownorm");
  launch_inv_rownorm(x.data_ptr(), inv.data_ptr<float>(), x.size(0),
                     cur_stream());
}

void rowscale(torch::Tensor x, torch::Tensor inv, torch::Tensor out) {
  CHK_CUDA(x); CHK_CONTIG(x); CHK_DT(x, torch::kBFloat16);
  CHK_CONTIG(out); CHK_DT(out, torch::kBFloat16);
  TORCH_CHECK(x.size(1) == 128 && out.sizes() == x.sizes() &&
              inv.numel() == x.size(0), "rowscale");
  launch_rowscale(x.data_ptr(), inv.data_ptr<float>(), out.data_ptr(),
                  x.size(0), cur_stream());
}

void angular_fwd(torch::Tensor ucv, torch::Tensor uw, torch::Tensor label,
                 torch::Tensor out, torch::Tensor cos_out, double cos_m,
                 double sin_m, double s) {
  CHK_CUDA(ucv); CHK_CONTIG(ucv); CHK_DT(ucv, torch::kBFloat16);
  CHK_CONTIG(uw); CHK_DT(uw, torch::kBFloat16);
  CHK_DT(label, torch::kInt64);
  CHK_CONTIG(out); CHK_DT(out, torch::kBFloat16);
  CHK_CONTIG(cos_out); CHK_DT(cos_out, torch::kBFloat16);
  const long B = ucv.size(0), L = uw.size(0);
  TORCH_CHECK(ucv.size(1) == 128 && uw.size(1) == 128 &&
              out.size(0) == B && out.size(1) == L &&
              cos_out.sizes() == out.sizes() && label.numel() == B,
              "angular_fwd shapes");
  launch_angular_fwd(ucv.data_ptr(), uw.data_ptr(), label.data_ptr<long>(),
                     out.data_ptr(), cos_out.data_ptr(), B, L, (float)cos_m,
                     (float)sin_m, (float)s, cur_stream());
}

void angular_dcos(torch::Tensor dout, torch::Tensor cosm,
                  torch::Tensor label, torch::Tensor dcos, double cos_m,
                  double sin_m, double s) {
  CHK_CUDA(dout); CHK_CONTIG(dout); CHK_DT(dout, torch::kBFloat16);
  CHK_CONTIG(cosm); CHK_DT(cosm, torch::kBFloat16);
  CHK_CONTIG(dcos); CHK_DT(dcos, torch::kBFloat16);
  const long B = dout.size(0), L = dout.size(1);
  TORCH_CHECK(cosm.sizes() == dout.sizes() && dcos.sizes() == dout.sizes()
              && label.numel() == B, "angular_dcos shapes");
  launch_angular_dcos(dout.data_ptr(), cosm.data_ptr(),
                      label.data_ptr<long>(), dcos.data_ptr(), B, L,
                      (float)cos_m, (float)sin_m, (float)s, cur_stream());
}

void norm_project(torch::Tensor du, torch::Tensor u, torch::Tensor inv,
                  torch::Tensor out) {
  CHK_CUDA(du); CHK_CONTIG(du); CHK_DT(du, torch::kBFloat16);
  CHK_CONTIG(u); CHK_DT(u, torch::kBFloat16);
  CHK_CONTIG(out); CHK_DT(out, torch::kBFloat16);
  TORCH_CHECK(du.size(1) == 128 && u.sizes() == du.sizes() &&
              out.sizes() == du.sizes() && inv.numel() == du.size(0),
              "norm_project shapes");
  launch_norm_project(du.data_ptr(), u.data_ptr(), inv.data_ptr<float>(),
                      out.data_ptr(), du.size(0), cur_stream());
}

// K17: per-row (max value, argmax) over bf16 logits (torch.max ties)
void row_max_argmax(torch::Tensor x, torch::Tensor vals, torch::Tensor idx) {
  CHK_CUDA(x); CHK_CONTIG(x); CHK_DT(x, torch::kBFloat16);
  CHK_DT(vals, torch::kFloat32); CHK_DT(idx, torch::kInt64);
  const long B = x.size(0), L = x.size(1);
  TORCH_CHECK(vals.numel() == B && idx.numel() == B, "row_max_argmax");
  launch_row_max_argmax(x.data_ptr(), vals.data_ptr<float>(),
                        idx.data_ptr<long>(), B, L, cur_stream());
}

void transpose_w(torch::Tensor w, torch::Tensor wt) {
  CHK_CUDA(w); CHK_CONTIG(w); CHK_DT(w, torch::kBFloat16);
  CHK_CONTIG(wt); CHK_DT(wt, torch::kBFloat16);
  const long L = w.size(0);
  TORCH_CHECK(w.size(1) == 128 && wt.size(0) == 128 && wt.size(1) == L &&
              L % 8 == 0, "transpose_w shapes");
  launch_transpose_w(w.data_ptr(), wt.data_ptr(), L, cur_stream());
}

void slab_sum_bf16(torch::Tensor partials, torch::Tensor out) {
  CHK_CUDA(partials); CHK_CONTIG(partials);
  CHK_DT(partials, torch::kFloat32);
  CHK_CONTIG(out); CHK_DT(out, torch::kBFloat16);
  const long N = out.numel();
  const long S = partials.numel() / N;
  TORCH_CHECK(S * N == partials.numel() && N % 4 == 0, "slab shapes");
  launch_slab_sum_bf16(partials.data_ptr<float>(), out.data_ptr(), (int)S,
                       N, cur_stream());
}

void slab_sum_f32(torch::Tensor p, torch::Tensor out) {
  CHK_CUDA(p); CHK_CONTIG(p); CHK_DT(p, torch::kFloat32);
  CHK_CONTIG(out); CHK_DT(out, torch::kFloat32);
  const int N = (int)out.numel();
  const int S = (int)(p.numel() / N);
  TORCH_CHECK((long)S * N == p.numel(), "slab_sum_f32 shapes");
  launch_slab_sum_f32(p.data_ptr<float>(), out.data_ptr<float>(), S, N,
                      cur_stream());
}

// dx = dz @ w2^T with w2 = [KP, EP=128] (the re-transposed combiner weight)
void dgrad2(torch::Tensor dz, torch::Tensor w2, torch::Tensor dx) {
  CHK_CUDA(dz); CHK_CONTIG(dz); CHK_DT(dz, torch::kBFloat16);
  CHK_CONTIG(w2); CHK_DT(w2, torch::kBFloat16);
  CHK_CONTIG(dx); CHK_DT(dx, torch::kBFloat16);
  const long M = dz.size(0), KP = w2.size(0);
  TORCH_CHECK(dz.size(1) == 128 && w2.size(1) == 128, "dgrad2 EP");
  TORCH_CHECK(dx.size(0) == M && dx.size(1) == KP && KP % 8 == 0,
              "dgrad2 shapes");
  launch_dgrad2(dz.data_ptr(), w2.data_ptr(), dx.data_ptr(), M, KP,
                cur_stream());
}

void colsum_bf16(torch::Tensor x, torch::Tensor out) {
  CHK_CUDA(x); CHK_CONTIG(x); CHK_DT(x, torch::kBFloat16);
  CHK_DT(out, torch::kFloat32);
  launch_colsum_bf16(x.data_ptr(), out.data_ptr<float>(), x.size(0),
                     x.size(1), cur_stream());
}

void wgrad(torch::Tensor x, torch::Tensor dz, torch::Tensor partials) {
  CHK_CUDA(x); CHK_CONTIG(x); CHK_DT(x, torch::kBFloat16);
  CHK_CONTIG(dz); CHK_DT(dz, torch::kBFloat16);
  CHK_DT(partials, torch::kFloat32); CHK_CONTIG(partials);
  const long M = x.size(0);
  const int KP = x.size(1), EP = dz.size(1);
  TORCH_CHECK(partials.size(1) == KP && partials.size(2) == EP, "partials");
  launch_wgrad(x.data_ptr(), dz.data_ptr(), partials.data_ptr<float>(), M,
               KP, EP, partials.size(0), cur_stream());
}

// bc_pow: float64 [2] DEVICE tensor = (beta1^t, beta2^t); advance it with
// adam_tick once per optimizer step (device-resident so hipGraph replays
// keep the bias correction advancing).
void adam_tick(torch::Tensor bc_pow, double b1, double b2) {
  CHK_CUDA(bc_pow); CHK_DT(bc_pow, torch::kFloat64);
  TORCH_CHECK(bc_pow.numel() == 2, "bc_pow must be [2]");
  launch_adam_tick((double*)bc_pow.data_ptr(), (float)b1, (float)b2,
                   cur_stream());
}

void adam_step_bf16(torch::Tensor p, torch::Tensor g, torch::Tensor master,
                    torch::Tensor m, torch::Tensor v, torch::Tensor bc_pow,
                    double lr, double b1, double b2, double eps, double wd) {
  CHK_CUDA(p); CHK_DT(p, torch::kBFloat16); CHK_DT(g, torch::kBFloat16);
  CHK_DT(master, torch::kFloat32); CHK_DT(bc_pow, torch::kFloat64);
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == master.numel(), "sizes");
  launch_adam_bf16(p.data_ptr(), g.data_ptr(), master.data_ptr<float>(),
                   m.data_ptr<float>(), v.data_ptr<float>(), p.numel(),
                   (const double*)bc_pow.data_ptr(), (float)lr, (float)b1,
                   (float)b2, (float)eps, (float)wd, cur_stream());
}

void adam_step_f32(torch::Tensor p, torch::Tensor g, torch::Tensor m,
                   torch::Tensor v, torch::Tensor bc_pow, double lr,
                   double b1, double b2, double eps, double wd) {
  CHK_CUDA(p); CHK_DT(p, torch::kFloat32); CHK_DT(g, torch::kFloat32);
  CHK_DT(bc_pow, torch::kFloat64);
  launch_adam_f32(p.data_ptr<float>(), g.data_ptr<float>(),
                  m.data_ptr<float>(), v.data_ptr<float>(), p.numel(),
                  (const double*)bc_pow.data_ptr(), (float)lr, (float)b1,
                  (float)b2, (float)eps, (float)wd, cur_stream());
}

// One launch for up to 8 small fp32 params (grads must be fp32 and
// contiguous); element-order within each tensor identical to the
// per-tensor kernel, so the update is bitwise-equal to 4 separate calls.
void adam_step_f32_multi(std::vector<torch::Tensor> ps,
                         std::vector<torch::Tensor> gs,
                         std::vector<torch::Tensor> ms,
                         std::vector<torch::Tensor> vs,
                         torch::Tensor bc_pow, double lr, double b1,
                         double b2, double eps, double wd) {
  const int n = (int)ps.size();
  TORCH_CHECK(n >= 1 && n <= 8, "adam_step_f32_multi: 1..8 tensors");
  TORCH_CHECK((int)gs.size() == n && (int)ms.size() == n &&
              (int)vs.size() == n, "adam_step_f32_multi: list sizes");
  CHK_DT(bc_pow, torch::kFloat64);
  float* pp[8]; const float* gp[8]; float* mp[8]; float* vp[8]; long ns[8];
  for (int t = 0; t < n; ++t) {
    CHK_CUDA(ps[t]); CHK_CONTIG(ps[t]); CHK_DT(ps[t], torch::kFloat32);
    CHK_CONTIG(gs[t]); CHK_DT(gs[t], torch::kFloat32);
    TORCH_CHECK(ps[t].numel() == gs[t].numel() &&
                ps[t].numel() == ms[t].numel() &&
                ps[t].numel() == vs[t].numel(), "sizes");
    pp[t] = ps[t].data_ptr<float>();
    gp[t] = gs[t].data_ptr<float>();
    mp[t] = ms[t].data_ptr<float>();
    vp[t] = vs[t].data_ptr<float>();
    ns[t] = ps[t].numel();
  }
  launch_adam_f32_multi(pp, gp, mp, vp, ns, n,
                        (const double*)bc_pow.data_ptr(), (float)lr,
                        (float)b1, (float)b2, (float)eps, (float)wd,
                        cur_stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("gather_concat_fwd", &gather_concat_fwd);
  m.def("gather_concat_bwd", &gather_concat_bwd);
  m.def("embed_scatter_sorted", &embed_scatter_sorted);
  m.def("group_by_index", &group_by_index);
  m.def("cast_clear_rows", &cast_clear_rows);
  m.def("combiner_fwd", &combiner_fwd);
  m.def("gather_combiner_fwd", &gather_combiner_fwd);
  m.def("wgrad_gather", &wgrad_gather);
  m.def("combiner_bwd", &combiner_bwd);
  m.def("attention_fwd", &attention_fwd);
  m.def("attention_bwd", &attention_bwd);
  m.def("logsoftmax_nll_fwd", &logsoftmax_nll_fwd);
  m.def("logsoftmax_nll_bwd", &logsoftmax_nll_bwd);
  m.def("wgrad", &wgrad);
  m.def("colsum_bf16", &colsum_bf16);
  m.def("head_fwd", &head_fwd);
  m.def("head_fwd_img", &head_fwd_img);
  m.def("swizzle_a", &swizzle_a);
  m.def("logsoftmax_nll_finalize", &logsoftmax_nll_finalize);
  m.def("lsm_partial", &lsm_partial);
  m.def("head_wgrad", &head_wgrad);
  m.def("head_dgrad", &head_dgrad);
  m.def("exclusive_scan", &exclusive_scan);
  m.def("cub_exclusive_scan", &cub_exclusive_scan);
  m.def("cub_scan_temp_bytes", &cub_scan_temp_bytes);
  m.def("head_bwd_prep", &head_bwd_prep);
  m.def("swizzle_cv", &swizzle_cv);
  m.def("row_max_argmax", &row_max_argmax);
  m.def("inv_rownorm", &inv_rownorm);
  m.def("rowscale", &rowscale);
  m.def("angular_fwd", &angular_fwd);
  m.def("angular_dcos", &angular_dcos);
  m.def("norm_project", &norm_project);
  m.def("head_bwd_dw", &head_bwd_dw);
  m.def("head_bwd_dcv", &head_bwd_dcv);
  m.def("head_bwd_dcv_rc", &head_bwd_dcv_rc);
  m.def("transpose_w", &transpose_w);
  m.def("dgrad2", &dgrad2);
  m.def("slab_sum_bf16", &slab_sum_bf16);
  m.def("slab_sum_f32", &slab_sum_f32);
  m.def("dgrad", &dgrad);
  m.def("adam_tick", &adam_tick);
  m.def("adam_step_bf16", &adam_step_bf16);
  m.def("adam_step_f32", &adam_step_f32);
  m.def("adam_step_f32_multi", &adam_step_f32_multi);
}
