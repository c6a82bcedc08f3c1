// Python bindings for the flowhip gfx950 HIP kernels (flowhip._C).

#include <torch/extension.h>

#include <vector>

#include <ATen/cuda/CUDAContext.h>
#include <c10/cuda/CUDAGuard.h>
#include <hip/hip_runtime.h>

// launchers defined in the .hip translation units
void flowhip_bgemm_nt_launch(const void* A, const void* B, void* C,
                             float alpha, int batch, int M, int N, int K,
                             int out_bf16, hipStream_t stream);
void flowhip_corr_lookup_fwd_launch(const void* const* levels, const int* Hs,
                                    const int* Ws, const float* coords,
                                    void* out, int BP, int P,
                                    int L, int radius, int cl, int ldc,
                                    int is_bf16, hipStream_t stream);
void flowhip_corr_lookup_bwd_launch(const void* gout, const float* coords,
                                    void* const* glevels, const int* Hs,
                                    const int* Ws, int BP, int P,
                                    int L, int radius, int cl,
                                    int is_bf16, int acc,
                                    hipStream_t stream);
void flowhip_convex_up_fwd_launch(const float* flow, const float* mask,
                                  float* out, int N, int H, int W, int factor,
                                  hipStream_t stream);
void flowhip_convex_up_bwd_launch(const float* gout, const float* flow,
                                  const float* mask, float* gflow,
                                  float* gmask, int N, int H, int W,
                                  int factor, hipStream_t stream);
void flowhip_nconv_fwd_launch(const float* data, const float* conf,
                              const float* weight, const float* bias,
                              float* out, float* cout, int N, int Ci, int Co,
                              int H, int W, int K, hipStream_t stream);
void flowhip_nconv_bwd_data_launch(const float* dnomin, const float* ddenom,
                                   const float* data, const float* conf,
                                   const float* weight, float* ddata,
                                   float* dconf, int N, int Ci, int Co, int H,
                                   int W, int K, hipStream_t stream);
void flowhip_nconv_wrw_launch(const float* dnomin, const float* ddenom,
                              const float* data, const float* conf,
                              float* dweight, int N, int Ci, int Co, int H,
                              int W, int K, hipStream_t stream);
int flowhip_nconv_tiled_nblocks(int N, int H, int W, int Ci);
bool flowhip_nconv_fwd_tiled_launch(const float* data, const float* conf,
                                    const float* weight, const float* bias,
                                    float* out, float* cout, int N, int Ci,
                                    int Co, int H, int W, int K,
                                    hipStream_t stream);
bool flowhip_nconv_bwd_data_tiled_launch(
    const float* dnomin, const float* ddenom, const float* data,
    const float* conf, const float* weight, float* ddata, float* dconf, int N,
    int Ci, int Co, int H, int W, int K, hipStream_t stream);
bool flowhip_nconv_wrw_tiled_launch(const float* dnomin, const float* ddenom,
                                    const float* data, const float* conf,
                                    float* partials, float* dweight, int N,
                                    int Ci, int Co, int H, int W, int K,
                                    hipStream_t stream);
int flowhip_instnorm_partial_rows(int N, int C, long P);
void flowhip_conv_gemm_pack_launch(const float* w, void* wpk, long total,
                                   int O, int I, int KH, int KW, int cpad,
                                   int flip, hipStream_t stream);
void flowhip_transpose_cast_launch(const void* in, void* out, int B, int M,
                                   int N, int in_bf16, hipStream_t stream);
void flowhip_conf_pool_fwd_launch(const float* data, const float* conf,
                                  float* data_ds, float* conf_ds,
                                  unsigned char* code, long total, int H,
                                  int W, int OH, int OW, hipStream_t stream);
void flowhip_conf_pool_bwd_launch(const float* gdata_ds,
                                  const float* gconf_ds,
                                  const unsigned char* code, float* gdata,
                                  float* gconf, long total_in, int H, int W,
                                  int OH, int OW, hipStream_t stream);
void flowhip_conv_gemm_fwd_launch(const void* x, const void* x2,
                                  const void* wpk, const float* bias,
                                  void* out, void* out2, const void* zpage,
                                  long Mtot, int HH, int WW, int srcH,
                                  int srcW, int sH, int sW, int ld_x,
                                  int ld_x2, int C1, int Cin, int Cout,
                                  int cpad, int KH, int KW, int padH,
                                  int padW, int osplit, int act, int smode,
                                  hipStream_t stream);
void flowhip_conv_gemm_wrw_launch(const void* dy, const void* x,
                                  const void* x2, float* partials, float* dw,
                                  float* dbias,
                                  const void* zpage, long Mtot, int HH,
                                  int WW, int srcH, int srcW, int sH, int sW,
                                  int ld_x, int ld_x2, int C1,
                                  int Cin, int Cout, int cpad, int KH,
                                  int KW, int padH, int padW, int nchunk,
                                  hipStream_t stream);
void flowhip_instnorm_cl_fwd_launch(const void* x, void* y, float* mean,
                                    float* rstd, float* partials, int N,
                                    int C, long P, float eps, int is_bf16,
                                    hipStream_t stream);
void flowhip_instnorm_cl_bwd_launch(const void* x, const void* dy,
                                    const float* mean, const float* rstd,
                                    float* gmean, float* gxmean,
                                    float* partials, void* dx, int N, int C,
                                    long P, int is_bf16, hipStream_t stream);
void flowhip_zero_inject_fwd_launch(const float* inp, float* out, long total,
                                    int ih, int iw, int oh, int ow, int sH,
                                    int sW, hipStream_t stream);
void flowhip_zero_inject_bwd_launch(const float* gout, float* dinp,
                                    long total, int ih, int iw, int oh,
                                    int ow, int sH, int sW,
                                    hipStream_t stream);
void flowhip_gru_gate1_fwd_launch(const void* zr, const void* h, void* z,
                                  void* rh, long total, int C, long P,
                                  int is_bf16, int cl, hipStream_t stream);
void flowhip_gru_gate1_bwd_launch(const void* dz, const void* drh,
                                  const void* zr, const void* h, void* dzr,
                                  void* dh, long total, int C, long P,
                                  int is_bf16, int cl, hipStream_t stream);
void flowhip_gru_gate2_fwd_launch(const void* qp, const void* z,
                                  const void* h, void* hnew, long total,
                                  int is_bf16, hipStream_t stream);
void flowhip_gru_gate2_bwd_launch(const void* dhnew, const void* qp,
                                  const void* z, const void* h, void* dqp,
                                  void* dz, void* dh, long total, int is_bf16,
                                  hipStream_t stream);
void flowhip_seq_loss_fwd_launch(const float* const* preds, int n,
                                 const float* gt, const float* valid,
                                 float* partials, float* out, long npix,
                                 long plane, float max_flow, float gamma,
                                 long numel_full, hipStream_t stream);
void flowhip_seq_loss_bwd_launch(const float* const* preds, float* const* dst,
                                 int n, const float* gt, const float* valid,
                                 const float* gloss, long npix, long plane,
                                 float max_flow, float gamma, long numel_full,
                                 hipStream_t stream);
void flowhip_nconv_bwd_prep_launch(const float* gout, const float* gcout,
                                   const float* out, const float* cout,
                                   const float* wsum, const float* bias,
                                   float* dnomin, float* ddenom, long total,
                                   long plane, int Co, float eps,
                                   hipStream_t stream);
bool flowhip_col_sum2_launch(const void* g, const void* x, float* out,
                             long M, int C, int nchunk,
                             hipStream_t stream);
void flowhip_frozen_bn_apply_launch(const void* x, void* y, const float* s,
                                    const float* t, long M, int C,
                                    hipStream_t stream);
void flowhip_plane_dot_sum_launch(const float* x, const float* y, float* out,
                                  long P, int B, int C, hipStream_t stream);
bool flowhip_col_sum_launch(const void* dy, float* out,
                            long M, int C, int nchunk, hipStream_t stream);
void flowhip_up2x_cat_fwd_launch(const float* low, const float* skip,
                                 float* out, long total, int C1, int C2,
                                 int H, int W, hipStream_t stream);
void flowhip_area_up2x_fwd_launch(const float* in, float* out, long total,
                                  int H, int W, hipStream_t stream);
void flowhip_area_up2x_bwd_launch(const float* gout, float* gin,
                                  long total_in, int H, int W,
                                  hipStream_t stream);
bool flowhip_packernel_fwd_launch(const float* f, float* k, int B, int C,
                                  int H, int W, int K, int dil, int norm,
                                  hipStream_t stream);
bool flowhip_packernel_bwd_launch(const float* f, const float* k,
                                  const float* dk, float* df, int B, int C,
                                  int H, int W, int K, int dil, int norm,
                                  hipStream_t stream);
bool flowhip_pacconv_fwd_launch(const float* x, const float* kr,
                                const float* w, const float* bias, float* out,
                                int B, int Ci, int Co, int H, int W, int OH,
                                int OW, int pH, int pW, int K, int dil,
                                int shared, hipStream_t stream);
bool flowhip_pacconv_bwd_launch(const float* dy, const float* x,
                                const float* kr, const float* w, float* dx,
                                float* dk, float* partials, float* dw,
                                int nchunk, int B, int Ci, int Co, int H,
                                int W, int OH, int OW, int pH, int pW, int K,
                                int dil, int shared, hipStream_t stream);
void flowhip_pacpool_fwd_launch(const float* x, const float* kr, float* out,
                                int B, int C, int KCH, int H, int W, int OH,
                                int OW, int K, int sH, int sW, int pH,
                                int pW, int dil, hipStream_t stream);
void flowhip_pacpool_bwd_launch(const float* dy, const float* x,
                                const float* kr, float* dx, float* dk, int B,
                                int C, int KCH, int H, int W, int OH, int OW,
                                int K, int sH, int sW, int pH, int pW,
                                int dil, hipStream_t stream);
bool flowhip_corr_pyramid_fwd_launch(const void* corr, void* l1, void* l2,
                                     void* l3, int BP, int H0, int W0,
                                     int nlev, int is_bf16,
                                     hipStream_t stream);
bool flowhip_corr_pyramid_fits(int H0, int W0, int is_bf16);
void flowhip_corr_pyramid_bwd_launch(const void* g0, const void* g1,
                                     const void* g2, const void* g3,
                                     void* dcorr, long total, int H0, int W0,
                                     int is_bf16, hipStream_t stream);

namespace {

torch::Tensor bgemm_nt(torch::Tensor a, torch::Tensor b, double alpha,
                       bool out_bf16) {
  TORCH_CHECK(a.is_cuda() && b.is_cuda(), "bgemm_nt: CUDA tensors required");
  TORCH_CHECK(a.dtype() == torch::kBFloat16 && b.dtype() == torch::kBFloat16,
              "bgemm_nt: bf16 operands required");
  TORCH_CHECK(a.dim() == 3 && b.dim() == 3, "bgemm_nt: (B,M,K) and (B,N,K)");
  TORCH_CHECK(a.is_contiguous() && b.is_contiguous());
  TORCH_CHECK(a.size(0) == b.size(0) && a.size(2) == b.size(2));
  TORCH_CHECK(a.size(2) % 64 == 0, "bgemm_nt: K must be a multiple of 64 "
              "(pad with zeros)");

  const int batch = a.size(0), M = a.size(1), N = b.size(1), K = a.size(2);
  auto c = torch::empty({batch, M, N},
                        a.options().dtype(out_bf16 ? torch::kBFloat16
                                                   : torch::kFloat32));
  const c10::cuda::CUDAGuard guard(a.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_bgemm_nt_launch(a.data_ptr(), b.data_ptr(), c.data_ptr(),
                          (float)alpha, batch, M, N, K, out_bf16 ? 1 : 0,
                          stream);
  return c;
}

torch::Tensor corr_lookup_fwd(std::vector<torch::Tensor> pyramid,
                              torch::Tensor coords, int64_t radius,
                              bool channels_last) {
  TORCH_CHECK(!pyramid.empty());
  TORCH_CHECK(coords.is_cuda() && coords.dtype() == torch::kFloat32 &&
              coords.is_contiguous());
  const int B = coords.size(0), H = coords.size(2), W = coords.size(3);
  const int P = H * W;
  const int L = (int)pyramid.size();
  const int K = 2 * (int)radius + 1;

  // channels-last: allocate with the channel count rounded up to 8 and
  // return a narrow view — the NHWC conv consumer reads whole 16-B pieces.
  // Output dtype follows the pyramid dtype (bf16-resident pyramid feeds
  // the bf16 motion-encoder conv with no cast pass).
  const bool bf = pyramid[0].dtype() == torch::kBFloat16;
  const auto odt = bf ? torch::kBFloat16 : torch::kFloat32;
  const long C = (long)L * K * K;
  const long C8 = channels_last ? (C + 7) / 8 * 8 : C;
  auto full = channels_last
                  ? torch::empty({B, C8, H, W},
                                 coords.options().dtype(odt)
                                     .memory_format(torch::MemoryFormat::ChannelsLast))
                  : torch::empty({B, C, H, W},
                                 coords.options().dtype(odt));
  auto out = (C8 != C) ? full.narrow(1, 0, C) : full;
  const c10::cuda::CUDAGuard guard(coords.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();

  const void* ptrs[4] = {nullptr, nullptr, nullptr, nullptr};
  int Hs[4] = {0, 0, 0, 0}, Ws[4] = {0, 0, 0, 0};
  TORCH_CHECK(L <= 4);
  for (int l = 0; l < L; ++l) {
    auto& lvl = pyramid[l];
    TORCH_CHECK(lvl.is_cuda() && lvl.is_contiguous() &&
                lvl.dtype() == (bf ? torch::kBFloat16 : torch::kFloat32),
                "corr_lookup: same-dtype contiguous pyramid levels required");
    TORCH_CHECK(lvl.size(0) == (long)B * P, "corr_lookup: level batch mismatch");
    ptrs[l] = lvl.data_ptr();
    Hs[l] = lvl.size(-2);
    Ws[l] = lvl.size(-1);
  }
  flowhip_corr_lookup_fwd_launch(
      ptrs, Hs, Ws, coords.data_ptr<float>(), full.data_ptr(), B * P, P, L,
      (int)radius, channels_last ? 1 : 0, (int)C8, bf ? 1 : 0, stream);
  return out;
}

std::vector<torch::Tensor> corr_lookup_bwd(torch::Tensor gout,
                                           torch::Tensor coords,
                                           int64_t radius,
                                           std::vector<std::vector<int64_t>>
                                               level_shapes,
                                           bool channels_last,
                                           bool levels_bf16,
                                           c10::optional<std::vector<torch::Tensor>>
                                               prev) {
  TORCH_CHECK(gout.is_cuda() &&
              gout.dtype() == (levels_bf16 ? torch::kBFloat16
                                           : torch::kFloat32),
              "corr_lookup_bwd: gout dtype must match the pyramid dtype");
  TORCH_CHECK(channels_last
                  ? gout.is_contiguous(torch::MemoryFormat::ChannelsLast)
                  : gout.is_contiguous());
  const int B = coords.size(0), H = coords.size(2), W = coords.size(3);
  const int P = H * W;
  const int L = (int)level_shapes.size();

  const c10::cuda::CUDAGuard guard(coords.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();

  // `prev` = the NEXT iteration's level grads (the pyramid is threaded
  // through the iteration loop as a chain): accumulate this lookup's
  // contribution into them IN KERNEL — no fresh zero-fill, no autograd
  // fan-in add_. Without prev: one flat zero-fill for all levels, plain
  // stores. Level-grad dtype matches the resident pyramid dtype.
  std::vector<torch::Tensor> grads;
  grads.reserve(L);
  const bool acc = prev.has_value();
  if (acc) {
    TORCH_CHECK((int)prev->size() == L);
    for (int l = 0; l < L; ++l) {
      auto& g = (*prev)[l];
      TORCH_CHECK(g.is_cuda() && g.is_contiguous() &&
                  g.dtype() == gout.dtype());
      grads.push_back(g);
    }
  } else {
    int64_t total = 0;
    std::vector<int64_t> sizes(L);
    for (int l = 0; l < L; ++l) {
      int64_t n = 1;
      for (auto d : level_shapes[l]) n *= d;
      sizes[l] = n;
      total += n;
    }
    auto flat = torch::zeros(
        {total}, gout.options().dtype(levels_bf16 ? torch::kBFloat16
                                                  : torch::kFloat32));
    int64_t off = 0;
    for (int l = 0; l < L; ++l) {
      grads.push_back(flat.narrow(0, off, sizes[l]).view(level_shapes[l]));
      off += sizes[l];
    }
  }
  void* gptrs[4] = {nullptr, nullptr, nullptr, nullptr};
  int Hs[4] = {0, 0, 0, 0}, Ws[4] = {0, 0, 0, 0};
  TORCH_CHECK(L <= 4);
  for (int l = 0; l < L; ++l) {
    gptrs[l] = grads[l].data_ptr();
    Hs[l] = grads[l].size(-2);
    Ws[l] = grads[l].size(-1);
  }
  flowhip_corr_lookup_bwd_launch(
      gout.data_ptr(), coords.data_ptr<float>(), gptrs, Hs, Ws,
      B * P, P, L, (int)radius, channels_last ? 1 : 0,
      levels_bf16 ? 1 : 0, acc ? 1 : 0, stream);
  return grads;
}

std::vector<torch::Tensor> corr_pyramid_fwd(torch::Tensor corr,
                                            int64_t num_levels) {
  const bool bf = corr.dtype() == torch::kBFloat16;
  TORCH_CHECK(corr.is_cuda() && (bf || corr.dtype() == torch::kFloat32) &&
              corr.is_contiguous());
  TORCH_CHECK(corr.dim() == 4 && corr.size(1) == 1);
  const long BP = corr.size(0);
  const int H0 = corr.size(2), W0 = corr.size(3);
  int H = H0, W = W0;
  std::vector<torch::Tensor> levels;
  for (int l = 1; l < num_levels; ++l) {
    H /= 2; W /= 2;
    levels.push_back(torch::empty({BP, 1, H, W}, corr.options()));
  }
  const c10::cuda::CUDAGuard guard(corr.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  void* p1 = levels.size() > 0 ? levels[0].data_ptr() : nullptr;
  void* p2 = levels.size() > 1 ? levels[1].data_ptr() : nullptr;
  void* p3 = levels.size() > 2 ? levels[2].data_ptr() : nullptr;
  bool ok = flowhip_corr_pyramid_fwd_launch(
      corr.data_ptr(), p1, p2, p3, (int)BP, H0, W0, (int)num_levels,
      bf ? 1 : 0, stream);
  TORCH_CHECK(ok, "corr_pyramid_fwd: unsupported shape (fall back to torch)");
  return levels;
}

bool corr_pyramid_fits(int64_t H0, int64_t W0, bool bf16) {
  return flowhip_corr_pyramid_fits((int)H0, (int)W0, bf16 ? 1 : 0);
}

torch::Tensor corr_pyramid_bwd(std::vector<c10::optional<torch::Tensor>> grads,
                               std::vector<int64_t> corr_shape) {
  TORCH_CHECK(grads.size() >= 1 && grads.size() <= 4);
  const long BP = corr_shape[0];
  const int H0 = corr_shape[2], W0 = corr_shape[3];
  const void* g[4] = {nullptr, nullptr, nullptr, nullptr};
  torch::TensorOptions opts;
  torch::Device dev(torch::kCUDA);
  bool have = false, bf = false;
  for (size_t l = 0; l < grads.size(); ++l) {
    if (grads[l].has_value()) {
      auto& t = grads[l].value();
      const bool tb = t.dtype() == torch::kBFloat16;
      TORCH_CHECK(t.is_cuda() && t.is_contiguous() &&
                  (tb || t.dtype() == torch::kFloat32));
      if (have) TORCH_CHECK(tb == bf, "corr_pyramid_bwd: mixed grad dtypes");
      g[l] = t.data_ptr();
      opts = t.options();
      dev = t.device();
      have = true;
      bf = tb;
    }
  }
  TORCH_CHECK(have, "corr_pyramid_bwd: all grads missing");
  auto dcorr = torch::empty(corr_shape, opts);
  const c10::cuda::CUDAGuard guard(dev);
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_corr_pyramid_bwd_launch(g[0], g[1], g[2], g[3],
                                  dcorr.data_ptr(),
                                  (long)BP * H0 * W0, H0, W0, bf ? 1 : 0,
                                  stream);
  return dcorr;
}

torch::Tensor convex_up_fwd(torch::Tensor flow, torch::Tensor mask,
                            int64_t factor) {
  TORCH_CHECK(flow.is_cuda() && flow.dtype() == torch::kFloat32 &&
              flow.is_contiguous());
  TORCH_CHECK(mask.is_cuda() && mask.dtype() == torch::kFloat32 &&
              mask.is_contiguous());
  TORCH_CHECK(factor == 8, "convex_up: factor 8 only");
  const int N = flow.size(0), H = flow.size(2), W = flow.size(3);
  TORCH_CHECK(flow.size(1) == 2 && mask.size(1) == 9 * factor * factor);

  auto out = torch::empty({N, 2, factor * H, factor * W}, flow.options());
  const c10::cuda::CUDAGuard guard(flow.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_convex_up_fwd_launch(flow.data_ptr<float>(), mask.data_ptr<float>(),
                               out.data_ptr<float>(), N, H, W, (int)factor,
                               stream);
  return out;
}

std::vector<torch::Tensor> convex_up_bwd(torch::Tensor gout,
                                         torch::Tensor flow,
                                         torch::Tensor mask, int64_t factor) {
  TORCH_CHECK(gout.is_cuda() && gout.is_contiguous() &&
              gout.dtype() == torch::kFloat32);
  const int N = flow.size(0), H = flow.size(2), W = flow.size(3);
  auto gflow = torch::zeros_like(flow);
  auto gmask = torch::empty_like(mask);
  const c10::cuda::CUDAGuard guard(flow.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_convex_up_bwd_launch(gout.data_ptr<float>(), flow.data_ptr<float>(),
                               mask.data_ptr<float>(),
                               gflow.data_ptr<float>(),
                               gmask.data_ptr<float>(), N, H, W, (int)factor,
                               stream);
  return {gflow, gmask};
}

std::vector<torch::Tensor> nconv_fwd(torch::Tensor data, torch::Tensor conf,
                                     torch::Tensor weight,
                                     c10::optional<torch::Tensor> bias) {
  TORCH_CHECK(data.is_cuda() && data.dtype() == torch::kFloat32 &&
              data.is_contiguous());
  TORCH_CHECK(conf.is_cuda() && conf.is_contiguous() && conf.sizes() == data.sizes());
  TORCH_CHECK(weight.is_cuda() && weight.is_contiguous() &&
              weight.dtype() == torch::kFloat32);
  const int N = data.size(0), Ci = data.size(1), H = data.size(2),
            W = data.size(3);
  const int Co = weight.size(0), K = weight.size(2);
  TORCH_CHECK(weight.size(1) == Ci && weight.size(3) == K);
  TORCH_CHECK(Ci <= 8 && Co <= 8 && (K == 1 || K == 3 || K == 5),
              "nconv_fwd: NCUP configuration space only");

  auto out = torch::empty({N, Co, H, W}, data.options());
  auto cout = torch::empty({N, Co, H, W}, data.options());
  const float* bptr = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->is_cuda() && bias->is_contiguous() &&
                bias->numel() == Co);
    bptr = bias->data_ptr<float>();
  }
  const c10::cuda::CUDAGuard guard(data.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  if (!flowhip_nconv_fwd_tiled_launch(
          data.data_ptr<float>(), conf.data_ptr<float>(),
          weight.data_ptr<float>(), bptr, out.data_ptr<float>(),
          cout.data_ptr<float>(), N, Ci, Co, H, W, K, stream)) {
    flowhip_nconv_fwd_launch(data.data_ptr<float>(), conf.data_ptr<float>(),
                             weight.data_ptr<float>(), bptr,
                             out.data_ptr<float>(), cout.data_ptr<float>(), N,
                             Ci, Co, H, W, K, stream);
  }
  return {out, cout};
}

std::vector<torch::Tensor> nconv_bwd(torch::Tensor dnomin,
                                     torch::Tensor ddenom, torch::Tensor data,
                                     torch::Tensor conf,
                                     torch::Tensor weight) {
  for (auto* t : {&dnomin, &ddenom, &data, &conf, &weight}) {
    TORCH_CHECK(t->is_cuda() && t->is_contiguous() &&
                t->dtype() == torch::kFloat32);
  }
  const int N = data.size(0), Ci = data.size(1), H = data.size(2),
            W = data.size(3);
  const int Co = weight.size(0), K = weight.size(2);

  auto ddata = torch::empty_like(data);
  auto dconf = torch::empty_like(conf);
  auto dweight = torch::zeros_like(weight);
  const c10::cuda::CUDAGuard guard(data.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  if (!flowhip_nconv_bwd_data_tiled_launch(
          dnomin.data_ptr<float>(), ddenom.data_ptr<float>(),
          data.data_ptr<float>(), conf.data_ptr<float>(),
          weight.data_ptr<float>(), ddata.data_ptr<float>(),
          dconf.data_ptr<float>(), N, Ci, Co, H, W, K, stream)) {
    flowhip_nconv_bwd_data_launch(
        dnomin.data_ptr<float>(), ddenom.data_ptr<float>(),
        data.data_ptr<float>(), conf.data_ptr<float>(),
        weight.data_ptr<float>(), ddata.data_ptr<float>(),
        dconf.data_ptr<float>(), N, Ci, Co, H, W, K, stream);
  }
  bool wrw_done = false;
  if (K == 1 || K == 3 || K == 5) {
    const int nblocks = flowhip_nconv_tiled_nblocks(N, H, W, Ci);
    auto partials = torch::empty({nblocks, (long)Co * Ci * K * K},
                                 data.options());
    wrw_done = flowhip_nconv_wrw_tiled_launch(
        dnomin.data_ptr<float>(), ddenom.data_ptr<float>(),
        data.data_ptr<float>(), conf.data_ptr<float>(),
        partials.data_ptr<float>(), dweight.data_ptr<float>(), N, Ci, Co, H,
        W, K, stream);
  }
  if (!wrw_done) {
    flowhip_nconv_wrw_launch(
        dnomin.data_ptr<float>(), ddenom.data_ptr<float>(),
        data.data_ptr<float>(), conf.data_ptr<float>(),
        dweight.data_ptr<float>(), N, Ci, Co, H, W, K, stream);
  }
  return {ddata, dconf, dweight};
}

std::vector<torch::Tensor> nconv_bwd_prep(
    torch::Tensor gout, c10::optional<torch::Tensor> gcout, torch::Tensor out,
    torch::Tensor cout, torch::Tensor wsum, c10::optional<torch::Tensor> bias,
    double eps) {
  TORCH_CHECK(gout.is_cuda() && gout.is_contiguous() &&
              gout.dtype() == torch::kFloat32);
  const int Co = gout.size(1);
  const long plane = (long)gout.size(2) * gout.size(3);
  const long total = gout.numel();
  auto dnomin = torch::empty_like(gout);
  auto ddenom = torch::empty_like(gout);
  const c10::cuda::CUDAGuard guard(gout.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_nconv_bwd_prep_launch(
      gout.data_ptr<float>(),
      gcout.has_value() ? gcout->data_ptr<float>() : nullptr,
      out.data_ptr<float>(), cout.data_ptr<float>(), wsum.data_ptr<float>(),
      bias.has_value() ? bias->data_ptr<float>() : nullptr,
      dnomin.data_ptr<float>(), ddenom.data_ptr<float>(), total, plane, Co,
      (float)eps, stream);
  return {dnomin, ddenom};
}

torch::Tensor seq_loss_fwd(std::vector<torch::Tensor> preds,
                           torch::Tensor gt, torch::Tensor valid,
                           double gamma, double max_flow) {
  const int n = (int)preds.size();
  TORCH_CHECK(n >= 1 && n <= 32, "seq_loss: 1..32 predictions");
  TORCH_CHECK(gt.is_cuda() && gt.is_contiguous() &&
              gt.dtype() == torch::kFloat32);
  TORCH_CHECK(valid.is_cuda() && valid.is_contiguous());
  const long B = gt.size(0), H = gt.size(2), W = gt.size(3);
  const long plane = H * W, npix = B * plane;
  std::vector<const float*> ptrs(n);
  for (int i = 0; i < n; ++i) {
    TORCH_CHECK(preds[i].is_cuda() && preds[i].is_contiguous() &&
                preds[i].dtype() == torch::kFloat32 &&
                preds[i].sizes() == gt.sizes());
    ptrs[i] = preds[i].data_ptr<float>();
  }
  auto partials = torch::empty({1024, (long)n + 5},
                               gt.options().dtype(torch::kFloat32));
  auto out = torch::empty({5}, gt.options().dtype(torch::kFloat32));
  const c10::cuda::CUDAGuard guard(gt.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_seq_loss_fwd_launch(ptrs.data(), n, gt.data_ptr<float>(),
                              valid.data_ptr<float>(),
                              partials.data_ptr<float>(),
                              out.data_ptr<float>(), npix, plane,
                              (float)max_flow, (float)gamma, gt.numel(),
                              stream);
  return out;
}

std::vector<torch::Tensor> seq_loss_bwd(std::vector<torch::Tensor> preds,
                                        torch::Tensor gt, torch::Tensor valid,
                                        torch::Tensor gloss, double gamma,
                                        double max_flow) {
  const int n = (int)preds.size();
  const long B = gt.size(0), H = gt.size(2), W = gt.size(3);
  const long plane = H * W, npix = B * plane;
  std::vector<const float*> ptrs(n);
  std::vector<float*> gptrs(n);
  std::vector<torch::Tensor> grads;
  grads.reserve(n);
  for (int i = 0; i < n; ++i) {
    ptrs[i] = preds[i].data_ptr<float>();
    grads.push_back(torch::empty_like(preds[i]));
    gptrs[i] = grads[i].data_ptr<float>();
  }
  const c10::cuda::CUDAGuard guard(gt.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_seq_loss_bwd_launch(ptrs.data(), gptrs.data(), n,
                              gt.data_ptr<float>(), valid.data_ptr<float>(),
                              gloss.data_ptr<float>(), npix, plane,
                              (float)max_flow, (float)gamma, gt.numel(),
                              stream);
  return grads;
}

namespace gg {

bool is_cl(const torch::Tensor& t) {
  return t.is_contiguous(torch::MemoryFormat::ChannelsLast);
}

torch::Tensor make_like(const torch::Tensor& t, bool cl) {
  return cl ? torch::empty(t.sizes(), t.options(),
                           torch::MemoryFormat::ChannelsLast)
            : torch::empty(t.sizes(), t.options());
}

}  // namespace gg

std::vector<torch::Tensor> gru_gate1_fwd(torch::Tensor zr, torch::Tensor h) {
  TORCH_CHECK(zr.is_cuda() && h.is_cuda());
  TORCH_CHECK(zr.scalar_type() == h.scalar_type());
  const bool bf16 = zr.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || zr.scalar_type() == torch::kFloat32);
  const bool cl = gg::is_cl(zr);
  TORCH_CHECK(cl ? gg::is_cl(h) : h.is_contiguous(),
              "gru_gate1: layouts must match");
  const int C = h.size(1);
  const long P = (long)h.size(2) * h.size(3);
  TORCH_CHECK(zr.size(1) == 2 * C && zr.size(0) == h.size(0));
  auto z = gg::make_like(h, cl);
  auto rh = gg::make_like(h, cl);
  const c10::cuda::CUDAGuard guard(h.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_gru_gate1_fwd_launch(zr.data_ptr(), h.data_ptr(), z.data_ptr(),
                               rh.data_ptr(), h.numel(), C, P, bf16 ? 1 : 0,
                               cl ? 1 : 0, stream);
  return {z, rh};
}

std::vector<torch::Tensor> gru_gate1_bwd(c10::optional<torch::Tensor> dz,
                                         torch::Tensor drh, torch::Tensor zr,
                                         torch::Tensor h) {
  const bool bf16 = zr.scalar_type() == torch::kBFloat16;
  const bool cl = gg::is_cl(zr);
  const int C = h.size(1);
  const long P = (long)h.size(2) * h.size(3);
  auto dzr = gg::make_like(zr, cl);
  auto dh = gg::make_like(h, cl);
  const c10::cuda::CUDAGuard guard(h.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_gru_gate1_bwd_launch(
      dz.has_value() ? dz->data_ptr() : nullptr, drh.data_ptr(),
      zr.data_ptr(), h.data_ptr(), dzr.data_ptr(), dh.data_ptr(), h.numel(),
      C, P, bf16 ? 1 : 0, cl ? 1 : 0, stream);
  return {dzr, dh};
}

torch::Tensor gru_gate2_fwd(torch::Tensor qp, torch::Tensor z,
                            torch::Tensor h) {
  const bool bf16 = qp.scalar_type() == torch::kBFloat16;
  const bool cl = gg::is_cl(qp);
  auto hnew = gg::make_like(h, cl);
  const c10::cuda::CUDAGuard guard(h.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_gru_gate2_fwd_launch(qp.data_ptr(), z.data_ptr(), h.data_ptr(),
                               hnew.data_ptr(), h.numel(), bf16 ? 1 : 0,
                               stream);
  return hnew;
}

std::vector<torch::Tensor> gru_gate2_bwd(torch::Tensor dhnew,
                                         torch::Tensor qp, torch::Tensor z,
                                         torch::Tensor h) {
  const bool bf16 = qp.scalar_type() == torch::kBFloat16;
  const bool cl = gg::is_cl(qp);
  auto dqp = gg::make_like(qp, cl);
  auto dz = gg::make_like(z, cl);
  auto dh = gg::make_like(h, cl);
  const c10::cuda::CUDAGuard guard(h.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_gru_gate2_bwd_launch(dhnew.data_ptr(), qp.data_ptr(), z.data_ptr(),
                               h.data_ptr(), dqp.data_ptr(), dz.data_ptr(),
                               dh.data_ptr(), h.numel(), bf16 ? 1 : 0,
                               stream);
  return {dqp, dz, dh};
}

torch::Tensor zero_inject_fwd(torch::Tensor inp, int64_t sH, int64_t sW,
                              int64_t oh, int64_t ow) {
  TORCH_CHECK(inp.is_cuda() && inp.dtype() == torch::kFloat32);
  auto x = inp.contiguous();
  const long N = x.size(0), C = x.size(1);
  const int ih = x.size(2), iw = x.size(3);
  auto out = torch::empty({N, C, oh, ow}, x.options());
  const c10::cuda::CUDAGuard guard(x.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_zero_inject_fwd_launch(x.data_ptr<float>(), out.data_ptr<float>(),
                                 out.numel(), ih, iw, (int)oh, (int)ow,
                                 (int)sH, (int)sW, stream);
  return out;
}

torch::Tensor zero_inject_bwd(torch::Tensor gout, int64_t sH, int64_t sW,
                              int64_t ih, int64_t iw) {
  auto g = gout.contiguous();
  const long N = g.size(0), C = g.size(1);
  const int oh = g.size(2), ow = g.size(3);
  auto dinp = torch::empty({N, C, ih, iw}, g.options());
  const c10::cuda::CUDAGuard guard(g.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_zero_inject_bwd_launch(g.data_ptr<float>(), dinp.data_ptr<float>(),
                                 dinp.numel(), (int)ih, (int)iw, oh, ow,
                                 (int)sH, (int)sW, stream);
  return dinp;
}

std::vector<torch::Tensor> instnorm_cl_fwd(torch::Tensor x, double eps) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4);
  TORCH_CHECK(x.is_contiguous(torch::MemoryFormat::ChannelsLast),
              "instnorm_cl: channels_last input required");
  const bool bf16 = x.scalar_type() == torch::kBFloat16;
  TORCH_CHECK(bf16 || x.scalar_type() == torch::kFloat32);
  const int N = x.size(0), C = x.size(1);
  const long P = (long)x.size(2) * x.size(3);
  auto y = torch::empty(x.sizes(), x.options(),
                        torch::MemoryFormat::ChannelsLast);
  auto mean = torch::empty({(long)N, (long)C},
                           x.options().dtype(torch::kFloat32));
  auto rstd = torch::empty_like(mean);
  auto partials = torch::empty({(long)flowhip_instnorm_partial_rows(N, C, P),
                                64},
                               x.options().dtype(torch::kFloat32));
  const c10::cuda::CUDAGuard guard(x.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_instnorm_cl_fwd_launch(x.data_ptr(), y.data_ptr(),
                                 mean.data_ptr<float>(),
                                 rstd.data_ptr<float>(),
                                 partials.data_ptr<float>(), N, C, P,
                                 (float)eps, bf16 ? 1 : 0, stream);
  return {y, mean, rstd};
}

torch::Tensor instnorm_cl_bwd(torch::Tensor x, torch::Tensor dy,
                              torch::Tensor mean, torch::Tensor rstd) {
  const bool bf16 = x.scalar_type() == torch::kBFloat16;
  const int N = x.size(0), C = x.size(1);
  const long P = (long)x.size(2) * x.size(3);
  auto dx = torch::empty(x.sizes(), x.options(),
                         torch::MemoryFormat::ChannelsLast);
  auto gmean = torch::empty_like(mean);
  auto gxmean = torch::empty_like(mean);
  auto partials = torch::empty({(long)flowhip_instnorm_partial_rows(N, C, P),
                                64},
                               x.options().dtype(torch::kFloat32));
  const c10::cuda::CUDAGuard guard(x.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_instnorm_cl_bwd_launch(x.data_ptr(), dy.data_ptr(),
                                 mean.data_ptr<float>(),
                                 rstd.data_ptr<float>(),
                                 gmean.data_ptr<float>(),
                                 gxmean.data_ptr<float>(),
                                 partials.data_ptr<float>(), dx.data_ptr(),
                                 N, C, P, bf16 ? 1 : 0, stream);
  return dx;
}

std::vector<torch::Tensor> conf_pool_fwd(torch::Tensor data,
                                         torch::Tensor conf) {
  TORCH_CHECK(data.is_cuda() && data.is_contiguous() &&
              data.dtype() == torch::kFloat32);
  TORCH_CHECK(conf.is_cuda() && conf.is_contiguous() &&
              conf.sizes() == data.sizes());
  const long N = data.size(0), C = data.size(1);
  const int H = data.size(2), W = data.size(3);
  const int OH = H / 2, OW = W / 2;
  auto dds = torch::empty({N, C, (long)OH, (long)OW}, data.options());
  auto cds = torch::empty_like(dds);
  auto code = torch::empty({N, C, (long)OH, (long)OW},
                           data.options().dtype(torch::kUInt8));
  const c10::cuda::CUDAGuard guard(data.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_conf_pool_fwd_launch(data.data_ptr<float>(),
                               conf.data_ptr<float>(), dds.data_ptr<float>(),
                               cds.data_ptr<float>(),
                               code.data_ptr<unsigned char>(), dds.numel(),
                               H, W, OH, OW, stream);
  return {dds, cds, code};
}

std::vector<torch::Tensor> conf_pool_bwd(c10::optional<torch::Tensor> gdds,
                                         c10::optional<torch::Tensor> gcds,
                                         torch::Tensor code,
                                         std::vector<int64_t> in_shape) {
  TORCH_CHECK(gdds.has_value() || gcds.has_value());
  auto& any = gdds.has_value() ? gdds.value() : gcds.value();
  const int H = in_shape[2], W = in_shape[3];
  const int OH = code.size(2), OW = code.size(3);
  auto gdata = torch::empty(in_shape, any.options());
  auto gconf = torch::empty(in_shape, any.options());
  const c10::cuda::CUDAGuard guard(any.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_conf_pool_bwd_launch(
      gdds.has_value() ? gdds->data_ptr<float>() : nullptr,
      gcds.has_value() ? gcds->data_ptr<float>() : nullptr,
      code.data_ptr<unsigned char>(), gdata.data_ptr<float>(),
      gconf.data_ptr<float>(), gdata.numel(), H, W, OH, OW, stream);
  return {gdata, gconf};
}

torch::Tensor transpose_cast_bf16(torch::Tensor in) {
  const bool bf = in.dtype() == torch::kBFloat16;
  TORCH_CHECK(in.is_cuda() && in.dim() == 3 && in.is_contiguous() &&
              (bf || in.dtype() == torch::kFloat32));
  const int B = in.size(0), M = in.size(1), N = in.size(2);
  auto out = torch::empty({(long)B, (long)N, (long)M},
                          in.options().dtype(torch::kBFloat16));
  const c10::cuda::CUDAGuard guard(in.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_transpose_cast_launch(in.data_ptr(), out.data_ptr(), B, M,
                                N, bf ? 1 : 0, stream);
  return out;
}

static const void* cg_zero_page() {
  static void* p = nullptr;
  if (p == nullptr) {
    hipMalloc(&p, 256);
    hipMemset(p, 0, 256);
  }
  return p;
}

// x: (N, C, H, W) with channels-last-like strides (stride(1)==1); a
// channel-narrowed view is allowed (ld_x = stride at dim 3 >= C).
static void cg_check_x(const torch::Tensor& x, int& ld) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
              x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(x.stride(1) == 1, "conv_gemm: channels-last layout required");
  ld = (int)x.stride(3);
  TORCH_CHECK(ld >= x.size(1) && x.stride(2) == (long)ld * x.size(3) &&
              x.stride(0) == (long)ld * x.size(3) * x.size(2));
}

torch::Tensor conv_gemm_pack(torch::Tensor w, bool flip) {
  TORCH_CHECK(w.is_cuda() && w.is_contiguous() &&
              w.scalar_type() == torch::kFloat32 && w.dim() == 4);
  const int O = w.size(0), I = w.size(1), KH = w.size(2), KW = w.size(3);
  const int R0 = flip ? I : O;
  const int cpad = ((flip ? O : I) + 63) / 64 * 64;
  auto wpk = torch::empty({(long)KH * KW, (long)R0, (long)cpad},
                          w.options().dtype(torch::kBFloat16));
  const c10::cuda::CUDAGuard guard(w.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_conv_gemm_pack_launch(w.data_ptr<float>(), wpk.data_ptr(),
                                wpk.numel(), O, I, KH, KW, cpad,
                                flip ? 1 : 0, stream);
  return wpk;
}

std::vector<torch::Tensor> conv_gemm_fwd2(
    torch::Tensor x, c10::optional<torch::Tensor> x2, torch::Tensor wpk,
    c10::optional<torch::Tensor> bias, int64_t Cout, int64_t KH, int64_t KW,
    int64_t osplit, int64_t act, int64_t sH, int64_t sW, int64_t smode,
    int64_t outH, int64_t outW) {
  int ld_x, ld_x2 = 0;
  cg_check_x(x, ld_x);
  TORCH_CHECK(wpk.is_cuda() && wpk.is_contiguous() &&
              wpk.scalar_type() == torch::kBFloat16 && wpk.dim() == 3);
  const int N = x.size(0), C1 = x.size(1), H = x.size(2), W = x.size(3);
  int Cin = C1;
  const void* x2p = nullptr;
  if (x2.has_value()) {
    int l2;
    cg_check_x(x2.value(), l2);
    TORCH_CHECK(x2->size(0) == N && x2->size(2) == H && x2->size(3) == W);
    TORCH_CHECK(C1 % 64 == 0, "conv_gemm cat2: first input must be a "
                "multiple of 64 channels");
    ld_x2 = l2;
    Cin = C1 + (int)x2->size(1);
    x2p = x2->data_ptr();
  }
  const int cpad = wpk.size(2);
  TORCH_CHECK(wpk.size(0) == KH * KW && wpk.size(1) == Cout);
  TORCH_CHECK(smode >= 0 && smode <= 2 && sH >= 1 && sW >= 1);
  // output spatial dims: same-pad for smode 0/1; smode 2 (strided
  // transposed conv, = backward-data of smode 1) takes them explicitly
  int OH = H, OW = W;
  if (smode == 1) {
    OH = (int)((H + 2 * (KH / 2) - KH) / sH + 1);
    OW = (int)((W + 2 * (KW / 2) - KW) / sW + 1);
  } else if (smode == 2) {
    TORCH_CHECK(outH > 0 && outW > 0, "conv_gemm smode=2 needs outH/outW");
    OH = (int)outH; OW = (int)outW;
  }
  const long Mtot = (long)N * OH * OW;
  const float* bptr = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->is_contiguous() &&
                bias->scalar_type() == torch::kFloat32 &&
                bias->numel() == Cout);
    bptr = bias->data_ptr<float>();
  }
  const c10::cuda::CUDAGuard guard(x.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  if (osplit <= 0 || osplit >= Cout) {
    auto out = torch::empty({(long)N, Cout, (long)OH, (long)OW}, x.options(),
                            torch::MemoryFormat::ChannelsLast);
    flowhip_conv_gemm_fwd_launch(x.data_ptr(), x2p, wpk.data_ptr(), bptr,
                                 out.data_ptr(), nullptr, cg_zero_page(),
                                 Mtot, OH, OW, H, W, (int)sH, (int)sW, ld_x,
                                 ld_x2, C1, Cin, (int)Cout,
                                 cpad, (int)KH, (int)KW, (int)KH / 2,
                                 (int)KW / 2, (int)Cout, (int)act,
                                 (int)smode, stream);
    return {out};
  }
  TORCH_CHECK(smode == 0, "conv_gemm: osplit only with smode 0");
  auto out = torch::empty({(long)N, osplit, (long)H, (long)W}, x.options(),
                          torch::MemoryFormat::ChannelsLast);
  auto out2 = torch::empty({(long)N, Cout - osplit, (long)H, (long)W},
                           x.options(), torch::MemoryFormat::ChannelsLast);
  flowhip_conv_gemm_fwd_launch(x.data_ptr(), x2p, wpk.data_ptr(), bptr,
                               out.data_ptr(), out2.data_ptr(),
                               cg_zero_page(), Mtot, H, W, H, W, 1, 1, ld_x,
                               ld_x2, C1,
                               Cin, (int)Cout, cpad, (int)KH, (int)KW,
                               (int)KH / 2, (int)KW / 2, (int)osplit,
                               (int)act, 0, stream);
  return {out, out2};
}

torch::Tensor conv_gemm_fwd(torch::Tensor x, torch::Tensor wpk,
                            c10::optional<torch::Tensor> bias, int64_t Cout,
                            int64_t KH, int64_t KW, int64_t act) {
  return conv_gemm_fwd2(x, c10::nullopt, wpk, bias, Cout, KH, KW, 0,
                        act, 1, 1, 0, 0, 0)[0];
}

std::vector<torch::Tensor> conv_gemm_wrw(torch::Tensor dy, torch::Tensor x,
                                         c10::optional<torch::Tensor> x2,
                                         int64_t KH, int64_t KW, int64_t sH,
                                         int64_t sW, bool want_bias) {
  int ld_x, ld_x2 = 0;
  cg_check_x(x, ld_x);
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == torch::kBFloat16 &&
              dy.stride(1) == 1 && dy.stride(3) == dy.size(1),
              "conv_gemm_wrw: dy must be channels-last contiguous");
  const int N = x.size(0), C1 = x.size(1), H = x.size(2), W = x.size(3);
  const int OH = dy.size(2), OW = dy.size(3);
  TORCH_CHECK(dy.size(0) == N);
  int Cin = C1;
  const void* x2p = nullptr;
  if (x2.has_value()) {
    int l2;
    cg_check_x(x2.value(), l2);
    ld_x2 = l2;
    Cin = C1 + (int)x2->size(1);
    x2p = x2->data_ptr();
  }
  const int Cout = dy.size(1);
  const long Mtot = (long)N * OH * OW;
  const int cpad = (int)((Cin + 63) / 64) * 64;
  const int tiles_o = (Cout + 63) / 64;
  // split M only as much as needed to fill the chip (~2 blocks/CU);
  // more chunks = more fp32 partial traffic + a longer reduce
  const int base_tiles = tiles_o * ((cpad + 63) / 64) * (int)(KH * KW);
  int nchunk = (512 + base_tiles - 1) / base_tiles;
  if (nchunk < 1) nchunk = 1;
  if (nchunk > 64) nchunk = 64;  // encoder shapes: few tiles, huge M
  // never more chunks than 64-row m-tiles
  const long mtiles = (Mtot + 63) / 64;
  if (nchunk > mtiles) nchunk = (int)mtiles;
  // bias partials (nchunk, tiles_o*64) ride at the tail of the same
  // buffer; dbias is written fully by the reduce kernel — no zero-fill
  auto partials = torch::empty(
      {(long)nchunk * KH * KW * tiles_o * 64 * cpad +
       (want_bias ? (long)nchunk * tiles_o * 64 : 0)},
      x.options().dtype(torch::kFloat32));
  auto dw = torch::empty({(long)Cout, (long)Cin, KH, KW},
                         x.options().dtype(torch::kFloat32));
  torch::Tensor dbias;
  if (want_bias)
    dbias = torch::empty({(long)Cout}, x.options().dtype(torch::kFloat32));
  const c10::cuda::CUDAGuard guard(x.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_conv_gemm_wrw_launch(dy.data_ptr(), x.data_ptr(), x2p,
                               partials.data_ptr<float>(),
                               dw.data_ptr<float>(),
                               want_bias ? dbias.data_ptr<float>() : nullptr,
                               cg_zero_page(), Mtot,
                               OH, OW, H, W, (int)sH, (int)sW,
                               ld_x, ld_x2, C1, Cin, Cout, cpad, (int)KH,
                               (int)KW, (int)KH / 2, (int)KW / 2, nchunk,
                               stream);
  return {dw, dbias};
}

torch::Tensor col_sum_bf16(torch::Tensor dy) {
  TORCH_CHECK(dy.is_cuda() && dy.dim() == 4 &&
              dy.scalar_type() == torch::kBFloat16 && dy.stride(1) == 1 &&
              dy.stride(3) == dy.size(1));
  const int C = dy.size(1);
  const long M = dy.numel() / C;
  const int nchunk = (int)((M + 255) / 256);  // must cover ALL rows (kernel bounds come from the chunk index)
  auto out = torch::zeros({(long)C}, dy.options().dtype(torch::kFloat32));
  const c10::cuda::CUDAGuard guard(dy.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  bool ok = flowhip_col_sum_launch(dy.data_ptr(),
                                   out.data_ptr<float>(), M, C, nchunk,
                                   stream);
  TORCH_CHECK(ok, "col_sum_bf16: C must be a multiple of 8");
  return out;
}

torch::Tensor plane_sum_nchw(torch::Tensor x, c10::optional<torch::Tensor> y) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.is_contiguous() &&
              x.scalar_type() == torch::kFloat32);
  const float* yp = nullptr;
  if (y.has_value()) {
    TORCH_CHECK(y->sizes() == x.sizes() && y->is_contiguous() &&
                y->scalar_type() == torch::kFloat32);
    yp = y->data_ptr<float>();
  }
  const int B = x.size(0), C = x.size(1);
  const long P = (long)x.size(2) * x.size(3);
  auto out = torch::zeros({(long)C}, x.options());
  const c10::cuda::CUDAGuard guard(x.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_plane_dot_sum_launch(x.data_ptr<float>(), yp,
                               out.data_ptr<float>(), P, B, C, stream);
  return out;
}

torch::Tensor frozen_bn_apply(torch::Tensor x, torch::Tensor s,
                              c10::optional<torch::Tensor> t) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 &&
              x.scalar_type() == torch::kBFloat16 && x.stride(1) == 1 &&
              x.stride(3) == x.size(1) && x.size(1) % 8 == 0);
  TORCH_CHECK(s.is_contiguous() && s.scalar_type() == torch::kFloat32 &&
              s.numel() == x.size(1));
  const float* tp = nullptr;
  if (t.has_value()) {
    TORCH_CHECK(t->is_contiguous() && t->scalar_type() == torch::kFloat32 &&
                t->numel() == x.size(1));
    tp = t->data_ptr<float>();
  }
  auto y = torch::empty_like(x);
  const c10::cuda::CUDAGuard guard(x.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_frozen_bn_apply_launch(x.data_ptr(), y.data_ptr(),
                                 s.data_ptr<float>(), tp,
                                 x.numel() / x.size(1), (int)x.size(1),
                                 stream);
  return y;
}

torch::Tensor col_sum2_bf16(torch::Tensor g, torch::Tensor x) {
  TORCH_CHECK(g.is_cuda() && g.dim() == 4 &&
              g.scalar_type() == torch::kBFloat16 && g.stride(1) == 1 &&
              g.stride(3) == g.size(1));
  TORCH_CHECK(x.sizes() == g.sizes() && x.strides() == g.strides() &&
              x.scalar_type() == torch::kBFloat16);
  const int C = g.size(1);
  const long M = g.numel() / C;
  const int nchunk = (int)((M + 255) / 256);
  auto out = torch::zeros({2, (long)C}, g.options().dtype(torch::kFloat32));
  const c10::cuda::CUDAGuard guard(g.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  bool ok = flowhip_col_sum2_launch(g.data_ptr(), x.data_ptr(),
                                    out.data_ptr<float>(), M, C, nchunk,
                                    stream);
  TORCH_CHECK(ok, "col_sum2_bf16: C must be a multiple of 8");
  return out;
}

torch::Tensor up2x_cat_fwd(torch::Tensor low, torch::Tensor skip) {
  TORCH_CHECK(low.is_cuda() && low.is_contiguous() &&
              low.dtype() == torch::kFloat32);
  TORCH_CHECK(skip.is_cuda() && skip.is_contiguous() &&
              skip.sizes()[0] == low.sizes()[0]);
  const long N = skip.size(0), C1 = low.size(1), C2 = skip.size(1);
  const int H = skip.size(2), W = skip.size(3);
  auto out = torch::empty({N, C1 + C2, (long)H, (long)W}, low.options());
  const c10::cuda::CUDAGuard guard(low.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_up2x_cat_fwd_launch(low.data_ptr<float>(), skip.data_ptr<float>(),
                              out.data_ptr<float>(), out.numel(), (int)C1,
                              (int)C2, H, W, stream);
  return out;
}

torch::Tensor area_up2x_fwd(torch::Tensor in) {
  TORCH_CHECK(in.is_cuda() && in.is_contiguous() && in.dim() == 4 &&
              in.dtype() == torch::kFloat32);
  const long N = in.size(0), C = in.size(1);
  const int H = in.size(2), W = in.size(3);
  auto out = torch::empty({N, C, (long)2 * H, (long)2 * W}, in.options());
  const c10::cuda::CUDAGuard guard(in.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_area_up2x_fwd_launch(in.data_ptr<float>(), out.data_ptr<float>(),
                               out.numel(), H, W, stream);
  return out;
}

torch::Tensor area_up2x_bwd(torch::Tensor gout) {
  auto g = gout.contiguous();
  const long N = g.size(0), C = g.size(1);
  const int H = g.size(2) / 2, W = g.size(3) / 2;
  auto gin = torch::empty({N, C, (long)H, (long)W}, g.options());
  const c10::cuda::CUDAGuard guard(g.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_area_up2x_bwd_launch(g.data_ptr<float>(), gin.data_ptr<float>(),
                               gin.numel(), H, W, stream);
  return gin;
}

torch::Tensor packernel_fwd(torch::Tensor f, int64_t K, int64_t dil,
                            bool normalize) {
  TORCH_CHECK(f.is_cuda() && f.is_contiguous() && f.dim() == 4 &&
              f.dtype() == torch::kFloat32);
  const int B = f.size(0), C = f.size(1), H = f.size(2), W = f.size(3);
  auto k = torch::empty({(long)B, K * K, (long)H, (long)W}, f.options());
  const c10::cuda::CUDAGuard guard(f.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  bool ok = flowhip_packernel_fwd_launch(f.data_ptr<float>(),
                                         k.data_ptr<float>(), B, C, H, W,
                                         (int)K, (int)dil, normalize ? 1 : 0,
                                         stream);
  TORCH_CHECK(ok, "packernel: unsupported K");
  return k;
}

torch::Tensor packernel_bwd(torch::Tensor f, torch::Tensor k,
                            torch::Tensor dk, int64_t K, int64_t dil,
                            bool normalize) {
  const int B = f.size(0), C = f.size(1), H = f.size(2), W = f.size(3);
  auto df = torch::empty_like(f);
  const c10::cuda::CUDAGuard guard(f.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  bool ok = flowhip_packernel_bwd_launch(
      f.data_ptr<float>(), k.data_ptr<float>(), dk.data_ptr<float>(),
      df.data_ptr<float>(), B, C, H, W, (int)K, (int)dil, normalize ? 1 : 0,
      stream);
  TORCH_CHECK(ok);
  return df;
}

torch::Tensor pacconv_fwd(torch::Tensor x, torch::Tensor kr,
                          torch::Tensor w, c10::optional<torch::Tensor> bias,
                          int64_t pH, int64_t pW, int64_t dil, bool shared) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.dtype() == torch::kFloat32);
  TORCH_CHECK(kr.is_contiguous() && w.is_contiguous());
  const int B = x.size(0), Ci = x.size(1), H = x.size(2), W = x.size(3);
  const int K = w.size(-1);
  const int Co = shared ? Ci : (int)w.size(0);
  const int OH = H + 2 * (int)pH - ((K - 1) * (int)dil);
  const int OW = W + 2 * (int)pW - ((K - 1) * (int)dil);
  TORCH_CHECK(kr.size(-2) == OH && kr.size(-1) == OW,
              "pacconv: kernel grid must match the output grid");
  auto out = torch::empty({(long)B, (long)Co, (long)OH, (long)OW},
                          x.options());
  const c10::cuda::CUDAGuard guard(x.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  bool ok = flowhip_pacconv_fwd_launch(
      x.data_ptr<float>(), kr.data_ptr<float>(), w.data_ptr<float>(),
      bias.has_value() ? bias->data_ptr<float>() : nullptr,
      out.data_ptr<float>(), B, Ci, Co, H, W, OH, OW, (int)pH, (int)pW, K,
      (int)dil, shared ? 1 : 0, stream);
  TORCH_CHECK(ok, "pacconv: unsupported K");
  return out;
}

std::vector<torch::Tensor> pacconv_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor kr, torch::Tensor w,
                                       int64_t pH, int64_t pW, int64_t dil,
                                       bool shared) {
  const int B = x.size(0), Ci = x.size(1), H = x.size(2), W = x.size(3);
  const int K = w.size(-1);
  const int Co = shared ? Ci : (int)w.size(0);
  const int OH = dy.size(2), OW = dy.size(3);
  const int K2 = K * K;
  const int nw = shared ? K2 : Co * Ci * K2;
  TORCH_CHECK((long)nw * 4 <= 65536, "pacconv dw: weight too large for the "
              "LDS-accumulated backward");
  const int nchunk = 512;
  auto dx = torch::empty_like(x);
  auto dk = torch::empty_like(kr);
  auto partials = torch::empty({(long)nchunk * nw},
                               x.options().dtype(torch::kFloat32));
  auto dw = torch::empty_like(w);
  const c10::cuda::CUDAGuard guard(x.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  bool ok = flowhip_pacconv_bwd_launch(
      dy.data_ptr<float>(), x.data_ptr<float>(), kr.data_ptr<float>(),
      w.data_ptr<float>(), dx.data_ptr<float>(), dk.data_ptr<float>(),
      partials.data_ptr<float>(), dw.data_ptr<float>(), nchunk, B, Ci, Co, H,
      W, OH, OW, (int)pH, (int)pW, K, (int)dil, shared ? 1 : 0, stream);
  TORCH_CHECK(ok);
  return {dx, dk, dw};
}

// kr: (B, KCH, K2, OH, OW) fp32; x: (B, C, H, W) fp32
torch::Tensor pacpool_fwd(torch::Tensor x, torch::Tensor kr, int64_t K,
                          int64_t sH, int64_t sW, int64_t pH, int64_t pW,
                          int64_t dil) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == torch::kFloat32);
  TORCH_CHECK(kr.is_cuda() && kr.is_contiguous() && kr.dim() == 5 &&
              kr.size(2) == K * K);
  const int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int KCH = kr.size(1), OH = kr.size(3), OW = kr.size(4);
  TORCH_CHECK(KCH == 1 || KCH == C);
  auto out = torch::empty({(long)B, (long)C, (long)OH, (long)OW},
                          x.options());
  const c10::cuda::CUDAGuard guard(x.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_pacpool_fwd_launch(x.data_ptr<float>(), kr.data_ptr<float>(),
                             out.data_ptr<float>(), B, C, KCH, H, W, OH, OW,
                             (int)K, (int)sH, (int)sW, (int)pH, (int)pW,
                             (int)dil, stream);
  return out;
}

std::vector<torch::Tensor> pacpool_bwd(torch::Tensor dy, torch::Tensor x,
                                       torch::Tensor kr, int64_t K,
                                       int64_t sH, int64_t sW, int64_t pH,
                                       int64_t pW, int64_t dil) {
  const int B = x.size(0), C = x.size(1), H = x.size(2), W = x.size(3);
  const int KCH = kr.size(1), OH = kr.size(3), OW = kr.size(4);
  auto dx = torch::zeros_like(x);
  auto dk = torch::empty_like(kr);
  const c10::cuda::CUDAGuard guard(x.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_pacpool_bwd_launch(dy.contiguous().data_ptr<float>(),
                             x.data_ptr<float>(), kr.data_ptr<float>(),
                             dx.data_ptr<float>(), dk.data_ptr<float>(), B,
                             C, KCH, H, W, OH, OW, (int)K, (int)sH, (int)sW,
                             (int)pH, (int)pW, (int)dil, stream);
  return {dx, dk};
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "flowhip gfx950 HIP kernels";
  m.def("bgemm_nt", &bgemm_nt,
        "C[b] = alpha * A[b] (M,K) @ B[b] (N,K)^T, bf16 in / fp32|bf16 out",
        py::arg("a"), py::arg("b"), py::arg("alpha"),
        py::arg("out_bf16") = false);
  m.def("corr_lookup_fwd", &corr_lookup_fwd,
        "fused multi-level correlation window lookup (fp32/bf16 levels)");
  m.def("corr_lookup_bwd", &corr_lookup_bwd,
        "backward of corr_lookup_fwd (pyramid grads; accumulates into "
        "`prev` when the pyramid is iteration-chained)",
        py::arg("gout"), py::arg("coords"), py::arg("radius"),
        py::arg("level_shapes"), py::arg("channels_last"),
        py::arg("levels_bf16") = false,
        py::arg("prev") = py::none());
  m.def("corr_pyramid_fits", &corr_pyramid_fits,
        "does the fused pyramid-build kernel support this map size/dtype");
  m.def("packernel_fwd", &packernel_fwd, "PAC gaussian adapting kernel");
  m.def("packernel_bwd", &packernel_bwd, "backward of packernel_fwd");
  m.def("pacconv_fwd", &pacconv_fwd, "pixel-adaptive convolution forward");
  m.def("pacconv_bwd", &pacconv_bwd, "pixel-adaptive conv backward (dx,dk,dw)");
  m.def("pacpool_fwd", &pacpool_fwd, "pixel-adaptive pooling forward");
  m.def("pacpool_bwd", &pacpool_bwd,
        "pixel-adaptive pooling backward (dx via atomic scatter, dk)");
  m.def("corr_pyramid_fwd", &corr_pyramid_fwd,
        "fused avg-pool pyramid build (levels 1..n-1)");
  m.def("corr_pyramid_bwd", &corr_pyramid_bwd,
        "fused pyramid backward combine -> dcorr");
  m.def("convex_up_fwd", &convex_up_fwd, "fused convex-combination upsample");
  m.def("convex_up_bwd", &convex_up_bwd, "backward of convex_up_fwd");
  m.def("nconv_fwd", &nconv_fwd,
        "fused normalized convolution forward (out, cout)");
  m.def("plane_sum_nchw", &plane_sum_nchw,
        "out[c] = sum_{b,h,w} x (or x*y), NCHW fp32",
        py::arg("x"), py::arg("y") = py::none());
  m.def("frozen_bn_apply", &frozen_bn_apply,
        "y = bf16(fp32(x)*s + t) per channel, channels-last");
  m.def("col_sum2_bf16", &col_sum2_bf16,
        "(sum_m g, sum_m g*x) per channel in one pass (frozen-BN backward)");
  m.def("col_sum_bf16", &col_sum_bf16,
        "(N,C,H,W) channels-last bf16 -> (C) fp32 bias-grad column sum");
  m.def("up2x_cat_fwd", &up2x_cat_fwd,
        "fused nearest-2x upsample + channel concat");
  m.def("area_up2x_fwd", &area_up2x_fwd, "exact-2x area upsample");
  m.def("area_up2x_bwd", &area_up2x_bwd, "backward of area_up2x");
  m.def("conf_pool_fwd", &conf_pool_fwd,
        "confidence-based 2x pooling forward (data_ds, conf_ds, argmax code)");
  m.def("conf_pool_bwd", &conf_pool_bwd, "backward of conf_pool");
  m.def("transpose_cast_bf16", &transpose_cast_bf16,
        "(B,M,N) fp32 -> (B,N,M) bf16 tiled transpose");
  m.def("conv_gemm_pack", &conv_gemm_pack,
        "one-kernel conv weight packing (fwd or flipped-bwd layout)");
  m.def("conv_gemm_fwd", &conv_gemm_fwd,
        "implicit-GEMM NHWC bf16 conv forward (also bwd-data with flipped "
        "packed weights)");
  m.def("conv_gemm_fwd2", &conv_gemm_fwd2,
        "conv forward over a virtually-concatenated pair of inputs and/or "
        "with a split output (bwd-data of a cat input); smode 1 = strided "
        "direct conv, smode 2 = strided transposed conv (bwd-data)",
        py::arg("x"), py::arg("x2"), py::arg("wpk"), py::arg("bias"),
        py::arg("Cout"), py::arg("KH"), py::arg("KW"), py::arg("osplit"),
        py::arg("act"), py::arg("sH") = 1, py::arg("sW") = 1,
        py::arg("smode") = 0, py::arg("outH") = 0, py::arg("outW") = 0);
  m.def("conv_gemm_wrw", &conv_gemm_wrw,
        "implicit-GEMM conv weight gradient (split-M + reduce); returns "
        "[dw, dbias] — dbias undefined unless want_bias",
        py::arg("dy"), py::arg("x"), py::arg("x2"), py::arg("KH"),
        py::arg("KW"), py::arg("sH") = 1, py::arg("sW") = 1,
        py::arg("want_bias") = false);
  m.def("instnorm_cl_fwd", &instnorm_cl_fwd,
        "channels-last InstanceNorm2d forward (y, mean, rstd)");
  m.def("instnorm_cl_bwd", &instnorm_cl_bwd,
        "channels-last InstanceNorm2d backward");
  m.def("zero_inject_fwd", &zero_inject_fwd, "sparse injection scatter");
  m.def("zero_inject_bwd", &zero_inject_bwd, "backward gather of inject");
  m.def("gru_gate1_fwd", &gru_gate1_fwd, "fused GRU z/r gates + r*h");
  m.def("gru_gate1_bwd", &gru_gate1_bwd, "backward of gru_gate1");
  m.def("gru_gate2_fwd", &gru_gate2_fwd, "fused GRU tanh + lerp update");
  m.def("gru_gate2_bwd", &gru_gate2_bwd, "backward of gru_gate2");
  m.def("seq_loss_fwd", &seq_loss_fwd,
        "fused sequence loss forward -> [loss, epe, 1px, 3px, 5px]");
  m.def("seq_loss_bwd", &seq_loss_bwd,
        "fused sequence loss backward -> per-prediction grads");
  m.def("nconv_bwd_prep", &nconv_bwd_prep,
        "fused elementwise preamble of nconv backward (dnomin, ddenom)");
  m.def("nconv_bwd", &nconv_bwd,
        "fused normalized convolution backward (ddata, dconf, dweight)");
}
