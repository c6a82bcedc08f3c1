// Python bindings for the flowhip gfx950 HIP kernels (flowhip._C).

#include <torch/extension.h>

#include <vector>

#include <ATen/cuda/CUDAContext.h>
#include <c10/cuda/CUDAGuard.h>
#include <hip/hip_runtime.h>

// launchers defined in the .hip translation units
void flowhip_bgemm_nt_launch(const void* A, const void* B, void* C,
                             float alpha, int batch, int M, int N, int K,
                             hipStream_t stream);
void flowhip_corr_lookup_fwd_launch(const float* level, const float* coords,
                                    float* out, int BP, int P, int Hl, int Wl,
                                    int l, int L, int radius, int cl,
                                    hipStream_t stream);
void flowhip_corr_lookup_bwd_launch(const float* gout, const float* coords,
                                    float* glevel, int BP, int P, int Hl,
                                    int Wl, int l, int L, int radius, int cl,
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
int flowhip_nconv_tiled_nblocks(int N, int H, int W);
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

namespace {

torch::Tensor bgemm_nt(torch::Tensor a, torch::Tensor b, double alpha) {
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
                        a.options().dtype(torch::kFloat32));
  const c10::cuda::CUDAGuard guard(a.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();
  flowhip_bgemm_nt_launch(a.data_ptr(), b.data_ptr(), c.data_ptr(),
                          (float)alpha, batch, M, N, K, stream);
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

  auto out = channels_last
                 ? torch::empty({B, (long)L * K * K, H, W},
                                coords.options().dtype(torch::kFloat32)
                                    .memory_format(torch::MemoryFormat::ChannelsLast))
                 : torch::empty({B, (long)L * K * K, H, W},
                                coords.options().dtype(torch::kFloat32));
  const c10::cuda::CUDAGuard guard(coords.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();

  for (int l = 0; l < L; ++l) {
    auto& lvl = pyramid[l];
    TORCH_CHECK(lvl.is_cuda() && lvl.is_contiguous() &&
                lvl.dtype() == torch::kFloat32,
                "corr_lookup: fp32 contiguous pyramid levels required");
    TORCH_CHECK(lvl.size(0) == (long)B * P, "corr_lookup: level batch mismatch");
    const int Hl = lvl.size(-2), Wl = lvl.size(-1);
    flowhip_corr_lookup_fwd_launch(
        lvl.data_ptr<float>(), coords.data_ptr<float>(), out.data_ptr<float>(),
        B * P, P, Hl, Wl, l, L, (int)radius, channels_last ? 1 : 0, stream);
  }
  return out;
}

std::vector<torch::Tensor> corr_lookup_bwd(torch::Tensor gout,
                                           torch::Tensor coords,
                                           int64_t radius,
                                           std::vector<std::vector<int64_t>>
                                               level_shapes,
                                           bool channels_last) {
  TORCH_CHECK(gout.is_cuda() && gout.dtype() == torch::kFloat32);
  TORCH_CHECK(channels_last
                  ? gout.is_contiguous(torch::MemoryFormat::ChannelsLast)
                  : gout.is_contiguous());
  const int B = coords.size(0), H = coords.size(2), W = coords.size(3);
  const int P = H * W;
  const int L = (int)level_shapes.size();

  const c10::cuda::CUDAGuard guard(coords.device());
  hipStream_t stream = at::cuda::getCurrentCUDAStream().stream();

  // one flat zero-fill for all levels (4x fewer fill kernels), sliced into
  // per-level views
  int64_t total = 0;
  std::vector<int64_t> sizes(L);
  for (int l = 0; l < L; ++l) {
    int64_t n = 1;
    for (auto d : level_shapes[l]) n *= d;
    sizes[l] = n;
    total += n;
  }
  auto flat = torch::zeros({total}, gout.options().dtype(torch::kFloat32));

  std::vector<torch::Tensor> grads;
  grads.reserve(L);
  int64_t off = 0;
  for (int l = 0; l < L; ++l) {
    auto g = flat.narrow(0, off, sizes[l]).view(level_shapes[l]);
    off += sizes[l];
    const int Hl = g.size(-2), Wl = g.size(-1);
    flowhip_corr_lookup_bwd_launch(
        gout.data_ptr<float>(), coords.data_ptr<float>(), g.data_ptr<float>(),
        B * P, P, Hl, Wl, l, L, (int)radius, channels_last ? 1 : 0, stream);
    grads.push_back(g);
  }
  return grads;
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
    const int nblocks = flowhip_nconv_tiled_nblocks(N, H, W);
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

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "flowhip gfx950 HIP kernels";
  m.def("bgemm_nt", &bgemm_nt,
        "C[b] = alpha * A[b] (M,K) @ B[b] (N,K)^T, bf16 in / fp32 out");
  m.def("corr_lookup_fwd", &corr_lookup_fwd,
        "fused multi-level correlation window lookup");
  m.def("corr_lookup_bwd", &corr_lookup_bwd,
        "backward of corr_lookup_fwd (pyramid grads)");
  m.def("convex_up_fwd", &convex_up_fwd, "fused convex-combination upsample");
  m.def("convex_up_bwd", &convex_up_bwd, "backward of convex_up_fwd");
  m.def("nconv_fwd", &nconv_fwd,
        "fused normalized convolution forward (out, cout)");
  m.def("nconv_bwd", &nconv_bwd,
        "fused normalized convolution backward (ddata, dconf, dweight)");
}
