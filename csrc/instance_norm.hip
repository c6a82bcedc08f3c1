// Channels-last InstanceNorm2d, fwd + bwd (affine=False,
// track_running_stats=False — the RAFT extractor configuration,
// reference extractor.py:25-27).
//
// PyTorch lowers InstanceNorm to batch_norm on a (1, N*C, H, W) view; on a
// channels-last tensor that view is non-contiguous, so every call pays an
// uncoalesced NCHW round-trip copy (~97 µs at (6,64,224,512), ~37 of them
// per training step — tprof6 in profiles/). This kernel reduces the (N,P,C)
// layout directly: lanes = consecutive channels (coalesced), waves split P.
//
//   fwd: y = (x - mu_{n,c}) * rstd_{n,c},  rstd = 1/sqrt(var + eps)
//   bwd: dx = rstd * (dy - mean_p(dy) - (x-mu)*rstd^2 * mean_p(dy*(x-mu)))
//
// One workgroup per (n, 64-channel block); two passes over P per kernel
// (reduce, then apply) — 3 HBM passes total, the memory-bound optimum for
// an unfused norm.

#include "common.h"

#define IN_THREADS 256
#define IN_CB 64  // channels per workgroup (= lanes per wave)

template <typename T>
__global__ __launch_bounds__(IN_THREADS) void instnorm_cl_fwd_kernel(
    const T* __restrict__ x, T* __restrict__ y, float* __restrict__ mean,
    float* __restrict__ rstd, int N, int C, long P, float eps) {
  const int n = blockIdx.x / ((C + IN_CB - 1) / IN_CB);
  const int cb = blockIdx.x % ((C + IN_CB - 1) / IN_CB);
  const int c = cb * IN_CB + (threadIdx.x & 63);
  const int wave = threadIdx.x >> 6;
  const bool cv = c < C;

  const T* xb = x + (long)n * P * C;
  float s1 = 0.f, s2 = 0.f;
  if (cv) {
    for (long p = wave; p < P; p += 4) {
      const float v = (float)xb[p * C + c];
      s1 += v;
      s2 += v * v;
    }
  }
  // cross-wave reduce: 4 partials per channel
  __shared__ float red1[4][IN_CB];
  __shared__ float red2[4][IN_CB];
  red1[wave][threadIdx.x & 63] = s1;
  red2[wave][threadIdx.x & 63] = s2;
  __syncthreads();
  __shared__ float smu[IN_CB], srs[IN_CB];
  if (threadIdx.x < IN_CB) {
    const float t1 = red1[0][threadIdx.x] + red1[1][threadIdx.x] +
                     red1[2][threadIdx.x] + red1[3][threadIdx.x];
    const float t2 = red2[0][threadIdx.x] + red2[1][threadIdx.x] +
                     red2[2][threadIdx.x] + red2[3][threadIdx.x];
    const float mu = t1 / (float)P;
    const float var = fmaxf(t2 / (float)P - mu * mu, 0.f);
    const float rs = rsqrtf(var + eps);
    smu[threadIdx.x] = mu;
    srs[threadIdx.x] = rs;
    const int cc = cb * IN_CB + (int)threadIdx.x;
    if (cc < C) {
      mean[(long)n * C + cc] = mu;
      rstd[(long)n * C + cc] = rs;
    }
  }
  __syncthreads();

  if (cv) {
    const float mu = smu[threadIdx.x & 63];
    const float rs = srs[threadIdx.x & 63];
    T* yb = y + (long)n * P * C;
    for (long p = wave; p < P; p += 4)
      yb[p * C + c] = (T)(((float)xb[p * C + c] - mu) * rs);
  }
}

template <typename T>
__global__ __launch_bounds__(IN_THREADS) void instnorm_cl_bwd_kernel(
    const T* __restrict__ x, const T* __restrict__ dy,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    T* __restrict__ dx, int N, int C, long P) {
  const int n = blockIdx.x / ((C + IN_CB - 1) / IN_CB);
  const int cb = blockIdx.x % ((C + IN_CB - 1) / IN_CB);
  const int c = cb * IN_CB + (threadIdx.x & 63);
  const int wave = threadIdx.x >> 6;
  const bool cv = c < C;

  const float mu = cv ? mean[(long)n * C + c] : 0.f;
  const float rs = cv ? rstd[(long)n * C + c] : 0.f;

  const T* xb = x + (long)n * P * C;
  const T* gb = dy + (long)n * P * C;
  float s1 = 0.f, s2 = 0.f;
  if (cv) {
    for (long p = wave; p < P; p += 4) {
      const float g = (float)gb[p * C + c];
      const float xc = (float)xb[p * C + c] - mu;
      s1 += g;
      s2 += g * xc;
    }
  }
  __shared__ float red1[4][IN_CB];
  __shared__ float red2[4][IN_CB];
  red1[wave][threadIdx.x & 63] = s1;
  red2[wave][threadIdx.x & 63] = s2;
  __syncthreads();
  __shared__ float sm1[IN_CB], sm2[IN_CB];
  if (threadIdx.x < IN_CB) {
    sm1[threadIdx.x] = (red1[0][threadIdx.x] + red1[1][threadIdx.x] +
                        red1[2][threadIdx.x] + red1[3][threadIdx.x]) /
                       (float)P;
    sm2[threadIdx.x] = (red2[0][threadIdx.x] + red2[1][threadIdx.x] +
                        red2[2][threadIdx.x] + red2[3][threadIdx.x]) /
                       (float)P;
  }
  __syncthreads();

  if (cv) {
    const float gmean = sm1[threadIdx.x & 63];
    const float gxmean = sm2[threadIdx.x & 63];
    T* db = dx + (long)n * P * C;
    for (long p = wave; p < P; p += 4) {
      const float g = (float)gb[p * C + c];
      const float xc = (float)xb[p * C + c] - mu;
      db[p * C + c] = (T)(rs * (g - gmean - xc * rs * rs * gxmean));
    }
  }
}

void flowhip_instnorm_cl_fwd_launch(const void* x, void* y, float* mean,
                                    float* rstd, int N, int C, long P,
                                    float eps, int is_bf16,
                                    hipStream_t stream) {
  const int grid = N * ((C + IN_CB - 1) / IN_CB);
  if (is_bf16)
    hipLaunchKernelGGL((instnorm_cl_fwd_kernel<__hip_bfloat16>), dim3(grid),
                       dim3(IN_THREADS), 0, stream,
                       (const __hip_bfloat16*)x, (__hip_bfloat16*)y, mean,
                       rstd, N, C, P, eps);
  else
    hipLaunchKernelGGL((instnorm_cl_fwd_kernel<float>), dim3(grid),
                       dim3(IN_THREADS), 0, stream, (const float*)x,
                       (float*)y, mean, rstd, N, C, P, eps);
}

void flowhip_instnorm_cl_bwd_launch(const void* x, const void* dy,
                                    const float* mean, const float* rstd,
                                    void* dx, int N, int C, long P,
                                    int is_bf16, hipStream_t stream) {
  const int grid = N * ((C + IN_CB - 1) / IN_CB);
  if (is_bf16)
    hipLaunchKernelGGL((instnorm_cl_bwd_kernel<__hip_bfloat16>), dim3(grid),
                       dim3(IN_THREADS), 0, stream,
                       (const __hip_bfloat16*)x, (const __hip_bfloat16*)dy,
                       mean, rstd, (__hip_bfloat16*)dx, N, C, P);
  else
    hipLaunchKernelGGL((instnorm_cl_bwd_kernel<float>), dim3(grid),
                       dim3(IN_THREADS), 0, stream, (const float*)x,
                       (const float*)dy, mean, rstd, (float*)dx, N, C, P);
}
