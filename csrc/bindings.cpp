// Python bindings for the flowhip gfx950 HIP kernels (flowhip._C).

#include <torch/extension.h>

#include <vector>

#include <ATen/hip/HIPContext.h>
#include <c10/hip/HIPGuard.h>

// launchers defined in the .hip translation units
void flowhip_bgemm_nt_launch(const void* A, const void* B, void* C,
                             float alpha, int batch, int M, int N, int K,
                             hipStream_t stream);
void flowhip_corr_lookup_fwd_launch(const float* level, const float* coords,
                                    float* out, int BP, int P, int Hl, int Wl,
                                    int l, int L, int radius,
                                    hipStream_t stream);
void flowhip_corr_lookup_bwd_launch(const float* gout, const float* coords,
                                    float* glevel, int BP, int P, int Hl,
                                    int Wl, int l, int L, int radius,
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
  const c10::hip::HIPGuard guard(a.device());
  auto stream = at::hip::getCurrentHIPStream();
  flowhip_bgemm_nt_launch(a.data_ptr(), b.data_ptr(), c.data_ptr(),
                          (float)alpha, batch, M, N, K, stream.stream());
  return c;
}

torch::Tensor corr_lookup_fwd(std::vector<torch::Tensor> pyramid,
                              torch::Tensor coords, int64_t radius) {
  TORCH_CHECK(!pyramid.empty());
  TORCH_CHECK(coords.is_cuda() && coords.dtype() == torch::kFloat32 &&
              coords.is_contiguous());
  const int B = coords.size(0), H = coords.size(2), W = coords.size(3);
  const int P = H * W;
  const int L = (int)pyramid.size();
  const int K = 2 * (int)radius + 1;

  auto out = torch::empty({B, (long)L * K * K, H, W},
                          coords.options().dtype(torch::kFloat32));
  const c10::hip::HIPGuard guard(coords.device());
  auto stream = at::hip::getCurrentHIPStream();

  for (int l = 0; l < L; ++l) {
    auto& lvl = pyramid[l];
    TORCH_CHECK(lvl.is_cuda() && lvl.is_contiguous() &&
                lvl.dtype() == torch::kFloat32,
                "corr_lookup: fp32 contiguous pyramid levels required");
    TORCH_CHECK(lvl.size(0) == (long)B * P, "corr_lookup: level batch mismatch");
    const int Hl = lvl.size(-2), Wl = lvl.size(-1);
    flowhip_corr_lookup_fwd_launch(
        lvl.data_ptr<float>(), coords.data_ptr<float>(), out.data_ptr<float>(),
        B * P, P, Hl, Wl, l, L, (int)radius, stream.stream());
  }
  return out;
}

std::vector<torch::Tensor> corr_lookup_bwd(torch::Tensor gout,
                                           torch::Tensor coords,
                                           int64_t radius,
                                           std::vector<std::vector<int64_t>>
                                               level_shapes) {
  TORCH_CHECK(gout.is_cuda() && gout.is_contiguous() &&
              gout.dtype() == torch::kFloat32);
  const int B = coords.size(0), H = coords.size(2), W = coords.size(3);
  const int P = H * W;
  const int L = (int)level_shapes.size();

  const c10::hip::HIPGuard guard(coords.device());
  auto stream = at::hip::getCurrentHIPStream();

  std::vector<torch::Tensor> grads;
  grads.reserve(L);
  for (int l = 0; l < L; ++l) {
    auto g = torch::zeros(level_shapes[l],
                          gout.options().dtype(torch::kFloat32));
    const int Hl = g.size(-2), Wl = g.size(-1);
    flowhip_corr_lookup_bwd_launch(
        gout.data_ptr<float>(), coords.data_ptr<float>(), g.data_ptr<float>(),
        B * P, P, Hl, Wl, l, L, (int)radius, stream.stream());
    grads.push_back(g);
  }
  return grads;
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
}
