// Fused ConvGRU gate elementwise (kernel #4 support, SURVEY.md §2.2;
// reference update.py:16-60). The convolutions stay on MIOpen (they are
// GEMM-shaped and well served); what this fuses is the gate math around
// them, which in eager mode is ~6 elementwise kernels per GRU pass per
// direction (sigmoid, chunk views, r*h, tanh, 3-op lerp) plus their
// backward — 2 fused kernels each way instead.
//
//   gate1:  z = sigmoid(zr[:, :C]);  r = sigmoid(zr[:, C:]);  rh = r * h
//   gate2:  hnew = (1-z) * h + z * tanh(qp)
//
// Backward recomputes the activations from the saved pre-activations
// (cheaper than saving z/r/t):
//   gate2: t = tanh(qp); dqp = dh_new*z*(1-t^2); dz = dh_new*(t-h);
//          dh = dh_new*(1-z)
//   gate1: dzr_z = dz_total*z*(1-z); dzr_r = (drh*h)*r*(1-r); dh += drh*r
//
// Supports fp32 and bf16 tensors (compute in fp32) in either NCHW or
// channels_last contiguous layout (all operands of one call share layout).

#include "common.h"

#define GG_THREADS 256

template <typename T>
__device__ inline float gg_ld(const T* p, long i) { return (float)p[i]; }
template <typename T>
__device__ inline void gg_st(T* p, long i, float v) { p[i] = (T)v; }

__device__ inline float gg_sigmoid(float x) { return 1.0f / (1.0f + expf(-x)); }

// index helpers: out tensors are (B, C, P); zr is (B, 2C, P).
// cl=1: channels-last ((b*P+p)*C + c); cl=0: NCHW ((b*C+c)*P + p).
struct GGIdx {
  long zr_z, zr_r;  // offsets of the z and r channels in zr for this element
};

template <int CL>
__device__ inline GGIdx gg_zr_idx(long idx, int C, long P) {
  GGIdx o;
  if (CL) {
    const long bp = idx / C;
    const int c = (int)(idx - bp * C);
    o.zr_z = bp * (2 * C) + c;
    o.zr_r = o.zr_z + C;
  } else {
    const long bc = idx / P;
    const long p = idx - bc * P;
    const long b = bc / C;
    const int c = (int)(bc - b * C);
    o.zr_z = ((b * 2 * C) + c) * P + p;
    o.zr_r = o.zr_z + (long)C * P;
  }
  return o;
}

template <typename T, int CL>
__global__ __launch_bounds__(GG_THREADS) void gru_gate1_fwd_kernel(
    const T* __restrict__ zr, const T* __restrict__ h, T* __restrict__ z,
    T* __restrict__ rh, long total, int C, long P) {
  for (long idx = (long)blockIdx.x * GG_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * GG_THREADS) {
    const GGIdx o = gg_zr_idx<CL>(idx, C, P);
    const float zv = gg_sigmoid(gg_ld(zr, o.zr_z));
    const float rv = gg_sigmoid(gg_ld(zr, o.zr_r));
    gg_st(z, idx, zv);
    gg_st(rh, idx, rv * gg_ld(h, idx));
  }
}

template <typename T, int CL>
__global__ __launch_bounds__(GG_THREADS) void gru_gate1_bwd_kernel(
    const T* __restrict__ dz, const T* __restrict__ drh,
    const T* __restrict__ zr, const T* __restrict__ h, T* __restrict__ dzr,
    T* __restrict__ dh, long total, int C, long P) {
  for (long idx = (long)blockIdx.x * GG_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * GG_THREADS) {
    const GGIdx o = gg_zr_idx<CL>(idx, C, P);
    const float zv = gg_sigmoid(gg_ld(zr, o.zr_z));
    const float rv = gg_sigmoid(gg_ld(zr, o.zr_r));
    const float gdz = dz ? gg_ld(dz, idx) : 0.0f;
    const float gdrh = gg_ld(drh, idx);
    gg_st(dzr, o.zr_z, gdz * zv * (1.0f - zv));
    gg_st(dzr, o.zr_r, gdrh * gg_ld(h, idx) * rv * (1.0f - rv));
    gg_st(dh, idx, gdrh * rv);
  }
}

template <typename T>
__global__ __launch_bounds__(GG_THREADS) void gru_gate2_fwd_kernel(
    const T* __restrict__ qp, const T* __restrict__ z,
    const T* __restrict__ h, T* __restrict__ hnew, long total) {
  for (long idx = (long)blockIdx.x * GG_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * GG_THREADS) {
    const float zv = gg_ld(z, idx);
    const float t = tanhf(gg_ld(qp, idx));
    gg_st(hnew, idx, (1.0f - zv) * gg_ld(h, idx) + zv * t);
  }
}

template <typename T>
__global__ __launch_bounds__(GG_THREADS) void gru_gate2_bwd_kernel(
    const T* __restrict__ dhnew, const T* __restrict__ qp,
    const T* __restrict__ z, const T* __restrict__ h, T* __restrict__ dqp,
    T* __restrict__ dz, T* __restrict__ dh, long total) {
  for (long idx = (long)blockIdx.x * GG_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * GG_THREADS) {
    const float g = gg_ld(dhnew, idx);
    const float zv = gg_ld(z, idx);
    const float t = tanhf(gg_ld(qp, idx));
    gg_st(dqp, idx, g * zv * (1.0f - t * t));
    gg_st(dz, idx, g * (t - gg_ld(h, idx)));
    gg_st(dh, idx, g * (1.0f - zv));
  }
}

static inline int gg_blocks(long total) {
  long b = (total + GG_THREADS - 1) / GG_THREADS;
  return (int)(b > 8192 ? 8192 : b);
}

#define GG_LAUNCH_T_CL(kernel, T, cl, ...)                                   \
  do {                                                                       \
    if (cl)                                                                  \
      hipLaunchKernelGGL((kernel<T, 1>), dim3(gg_blocks(total)),             \
                         dim3(GG_THREADS), 0, stream, __VA_ARGS__);          \
    else                                                                     \
      hipLaunchKernelGGL((kernel<T, 0>), dim3(gg_blocks(total)),             \
                         dim3(GG_THREADS), 0, stream, __VA_ARGS__);          \
  } while (0)

void flowhip_gru_gate1_fwd_launch(const void* zr, const void* h, void* z,
                                  void* rh, long total, int C, long P,
                                  int is_bf16, int cl, hipStream_t stream) {
  if (is_bf16)
    GG_LAUNCH_T_CL(gru_gate1_fwd_kernel, __hip_bfloat16, cl,
                   (const __hip_bfloat16*)zr, (const __hip_bfloat16*)h,
                   (__hip_bfloat16*)z, (__hip_bfloat16*)rh, total, C, P);
  else
    GG_LAUNCH_T_CL(gru_gate1_fwd_kernel, float, cl, (const float*)zr,
                   (const float*)h, (float*)z, (float*)rh, total, C, P);
}

void flowhip_gru_gate1_bwd_launch(const void* dz, const void* drh,
                                  const void* zr, const void* h, void* dzr,
                                  void* dh, long total, int C, long P,
                                  int is_bf16, int cl, hipStream_t stream) {
  if (is_bf16)
    GG_LAUNCH_T_CL(gru_gate1_bwd_kernel, __hip_bfloat16, cl,
                   (const __hip_bfloat16*)dz, (const __hip_bfloat16*)drh,
                   (const __hip_bfloat16*)zr, (const __hip_bfloat16*)h,
                   (__hip_bfloat16*)dzr, (__hip_bfloat16*)dh, total, C, P);
  else
    GG_LAUNCH_T_CL(gru_gate1_bwd_kernel, float, cl, (const float*)dz,
                   (const float*)drh, (const float*)zr, (const float*)h,
                   (float*)dzr, (float*)dh, total, C, P);
}

void flowhip_gru_gate2_fwd_launch(const void* qp, const void* z,
                                  const void* h, void* hnew, long total,
                                  int is_bf16, hipStream_t stream) {
  if (is_bf16)
    hipLaunchKernelGGL((gru_gate2_fwd_kernel<__hip_bfloat16>),
                       dim3(gg_blocks(total)), dim3(GG_THREADS), 0, stream,
                       (const __hip_bfloat16*)qp, (const __hip_bfloat16*)z,
                       (const __hip_bfloat16*)h, (__hip_bfloat16*)hnew,
                       total);
  else
    hipLaunchKernelGGL((gru_gate2_fwd_kernel<float>), dim3(gg_blocks(total)),
                       dim3(GG_THREADS), 0, stream, (const float*)qp,
                       (const float*)z, (const float*)h, (float*)hnew, total);
}

void flowhip_gru_gate2_bwd_launch(const void* dhnew, const void* qp,
                                  const void* z, const void* h, void* dqp,
                                  void* dz, void* dh, long total, int is_bf16,
                                  hipStream_t stream) {
  if (is_bf16)
    hipLaunchKernelGGL((gru_gate2_bwd_kernel<__hip_bfloat16>),
                       dim3(gg_blocks(total)), dim3(GG_THREADS), 0, stream,
                       (const __hip_bfloat16*)dhnew,
                       (const __hip_bfloat16*)qp, (const __hip_bfloat16*)z,
                       (const __hip_bfloat16*)h, (__hip_bfloat16*)dqp,
                       (__hip_bfloat16*)dz, (__hip_bfloat16*)dh, total);
  else
    hipLaunchKernelGGL((gru_gate2_bwd_kernel<float>), dim3(gg_blocks(total)),
                       dim3(GG_THREADS), 0, stream, (const float*)dhnew,
                       (const float*)qp, (const float*)z, (const float*)h,
                       (float*)dqp, (float*)dz, (float*)dh, total);
}
