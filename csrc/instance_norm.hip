// Channels-last InstanceNorm2d, fwd + bwd (affine=False,
// track_running_stats=False — the RAFT extractor configuration,
// reference extractor.py:25-27).
//
// PyTorch lowers InstanceNorm to batch_norm on a (1, N*C, H, W) view; on a
// channels-last tensor that view is non-contiguous, so every call pays an
// uncoalesced NCHW round-trip copy (~97 µs at (6,64,224,512), ~37 of them
// per training step — tprof6 in profiles/). These kernels reduce the
// (N,P,C) layout directly: lanes = consecutive channels (coalesced).
//
// Parallel structure (v2 — v1 used one workgroup per (n, c-block), i.e.
// 6-12 workgroups on a 256-CU chip, a 10x step regression): three stages,
//   1. partial: grid (N * CB * PCHUNKS), each block reduces a P-chunk ->
//      partials (n, cb, chunk, {s1,s2}, 64)
//   2. finalize: one small kernel -> mean/rstd (N, C)
//   3. apply: grid-stride elementwise normalize / backward-apply
// All passes are coalesced in the channels-last layout.
//
//   fwd: y = (x - mu_{n,c}) * rstd_{n,c},  rstd = 1/sqrt(var + eps)
//   bwd: dx = rstd * (dy - mean_p(dy) - (x-mu)*rstd^2 * mean_p(dy*(x-mu)))

#include "common.h"

#include <type_traits>

#define IN_THREADS 256
#define IN_CB 64      // channels per block column
#define IN_PCHUNK 1024  // pixels per partial-reduce block

// mode 0: s1 = x, s2 = x*x          (fwd statistics)
// mode 1: s1 = dy, s2 = dy*(x-mu)   (bwd reductions; a = dy)
//
// Thread layout (v3): each thread owns EIGHT consecutive channels and one
// p-stream — per-lane 16-byte bf16x8 loads instead of one 2-byte element
// (the v2 per-lane-per-channel form was load-issue-bound at 427 us for a
// 44 M-element plane; ~40 us is the bandwidth bound). Requires C % 8 == 0
// (all extractor widths); the module falls back to torch otherwise.
template <typename T, int MODE>
__global__ __launch_bounds__(IN_THREADS) void instnorm_partial_kernel(
    const T* __restrict__ x, const T* __restrict__ a,
    const float* __restrict__ mean, float* __restrict__ partials,
    int N, int C, long P, int nchunk) {
  const int ncb = (C + IN_CB - 1) / IN_CB;
  int b = blockIdx.x;
  const int chunk = b % nchunk; b /= nchunk;
  const int cb = b % ncb; b /= ncb;
  const int n = b;
  const int CBW = min(IN_CB, C - cb * IN_CB);  // channels in this block
  const int CG = CBW / 8;                      // 8-channel groups (<= 8)
  const int g = threadIdx.x % CG;              // this thread's group
  const int s = threadIdx.x / CG;              // p-stream
  const int S = IN_THREADS / CG;               // streams
  const int c0 = cb * IN_CB + g * 8;

  const long p0 = (long)chunk * IN_PCHUNK;
  const long p1 = min(p0 + IN_PCHUNK, P);

  float s1[8], s2[8], mu[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) { s1[j] = 0.f; s2[j] = 0.f; mu[j] = 0.f; }
  if (MODE == 1 && s < S) {
#pragma unroll
    for (int j = 0; j < 8; ++j) mu[j] = mean[(long)n * C + c0 + j];
  }

  if (s < S) {
    const T* xb = x + (long)n * P * C;
    const T* ab = (MODE == 1) ? a + (long)n * P * C : nullptr;
    // one 16-B load per tensor (bf16x8 / f32x4 pair)
    using Tv = typename std::conditional<sizeof(T) == 2, bf16x8, f32x4>::type;
    constexpr int VW = sizeof(T) == 2 ? 8 : 4;
    for (long p = p0 + s; p < p1; p += S) {
      Tv xv0 = *(const Tv*)(xb + p * C + c0);
      Tv xv1 = xv0;
      if (VW == 4) xv1 = *(const Tv*)(xb + p * C + c0 + 4);
      Tv av0 = xv0, av1 = xv1;
      if (MODE == 1) {
        av0 = *(const Tv*)(ab + p * C + c0);
        if (VW == 4) av1 = *(const Tv*)(ab + p * C + c0 + 4);
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float v = (float)(j < VW ? xv0[j % VW] : xv1[j % VW]);
        if (MODE == 0) {
          s1[j] += v;
          s2[j] += v * v;
        } else {
          const float gv = (float)(j < VW ? av0[j % VW] : av1[j % VW]);
          s1[j] += gv;
          s2[j] += gv * (v - mu[j]);
        }
      }
    }
  }

  // LDS reduce over the p-streams: lds[2][64 channels][stride]
  __shared__ float red[2 * IN_CB * 33];
  const int stride = 33;  // odd stride: no bank conflicts across s
  for (int j = 0; j < 8; ++j) {
    const int c = g * 8 + j;
    if (s < S && s < 32) {
      red[(c)*stride + s] = s1[j];
      red[(IN_CB + c) * stride + s] = s2[j];
    }
  }
  __syncthreads();
  // streams beyond 32 fold in (S can be 42 for CG=6... cap: accumulate)
  if (s >= 32 && s < S) {
    for (int j = 0; j < 8; ++j) {
      const int c = g * 8 + j;
      atomicAdd(&red[(c)*stride + (s & 31)], s1[j]);
      atomicAdd(&red[(IN_CB + c) * stride + (s & 31)], s2[j]);
    }
  }
  __syncthreads();
  if (threadIdx.x < (unsigned)CBW) {
    float t1 = 0.f, t2 = 0.f;
    const int smax = S < 32 ? S : 32;
    for (int ss = 0; ss < smax; ++ss) {
      t1 += red[threadIdx.x * stride + ss];
      t2 += red[(IN_CB + threadIdx.x) * stride + ss];
    }
    const long row = (((long)n * ncb + cb) * nchunk + chunk) * 2;
    partials[(row + 0) * IN_CB + threadIdx.x] = t1;
    partials[(row + 1) * IN_CB + threadIdx.x] = t2;
  }
}

// finalize fwd: mean/rstd from partials. grid (N*ncb), 64 threads.
__global__ __launch_bounds__(64) void instnorm_finalize_fwd_kernel(
    const float* __restrict__ partials, float* __restrict__ mean,
    float* __restrict__ rstd, int N, int C, long P, int nchunk, float eps) {
  const int ncb = (C + IN_CB - 1) / IN_CB;
  const int cb = blockIdx.x % ncb;
  const int n = blockIdx.x / ncb;
  const int c = cb * IN_CB + threadIdx.x;
  float s1 = 0.f, s2 = 0.f;
  for (int ch = 0; ch < nchunk; ++ch) {
    const long row = (((long)n * ncb + cb) * nchunk + ch) * 2;
    s1 += partials[(row + 0) * IN_CB + threadIdx.x];
    s2 += partials[(row + 1) * IN_CB + threadIdx.x];
  }
  if (c < C) {
    const float mu = s1 / (float)P;
    const float var = fmaxf(s2 / (float)P - mu * mu, 0.f);
    mean[(long)n * C + c] = mu;
    rstd[(long)n * C + c] = rsqrtf(var + eps);
  }
}

// finalize bwd: means of (dy, dy*(x-mu)). grid (N*ncb), 64 threads.
__global__ __launch_bounds__(64) void instnorm_finalize_bwd_kernel(
    const float* __restrict__ partials, float* __restrict__ gmean,
    float* __restrict__ gxmean, int N, int C, long P, int nchunk) {
  const int ncb = (C + IN_CB - 1) / IN_CB;
  const int cb = blockIdx.x % ncb;
  const int n = blockIdx.x / ncb;
  const int c = cb * IN_CB + threadIdx.x;
  float s1 = 0.f, s2 = 0.f;
  for (int ch = 0; ch < nchunk; ++ch) {
    const long row = (((long)n * ncb + cb) * nchunk + ch) * 2;
    s1 += partials[(row + 0) * IN_CB + threadIdx.x];
    s2 += partials[(row + 1) * IN_CB + threadIdx.x];
  }
  if (c < C) {
    gmean[(long)n * C + c] = s1 / (float)P;
    gxmean[(long)n * C + c] = s2 / (float)P;
  }
}

template <typename T>
__global__ __launch_bounds__(IN_THREADS) void instnorm_apply_fwd_kernel(
    const T* __restrict__ x, T* __restrict__ y,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    long total, int C, long P) {
  for (long idx = (long)blockIdx.x * IN_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * IN_THREADS) {
    const long n = idx / (P * C);
    const int c = (int)(idx % C);
    const long s = n * C + c;
    y[idx] = (T)(((float)x[idx] - mean[s]) * rstd[s]);
  }
}

template <typename T>
__global__ __launch_bounds__(IN_THREADS) void instnorm_apply_bwd_kernel(
    const T* __restrict__ x, const T* __restrict__ dy, T* __restrict__ dx,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gmean, const float* __restrict__ gxmean,
    long total, int C, long P) {
  for (long idx = (long)blockIdx.x * IN_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * IN_THREADS) {
    const long n = idx / (P * C);
    const int c = (int)(idx % C);
    const long s = n * C + c;
    const float rs = rstd[s];
    const float xc = (float)x[idx] - mean[s];
    dx[idx] = (T)(rs * ((float)dy[idx] - gmean[s] - xc * rs * rs * gxmean[s]));
  }
}

static inline int in_nchunk(long P) { return (int)((P + IN_PCHUNK - 1) / IN_PCHUNK); }

int flowhip_instnorm_partial_rows(int N, int C, long P) {
  return N * ((C + IN_CB - 1) / IN_CB) * in_nchunk(P) * 2;
}

template <typename T>
static void instnorm_fwd_t(const T* x, T* y, float* mean, float* rstd,
                           float* partials, int N, int C, long P, float eps,
                           hipStream_t stream) {
  const int ncb = (C + IN_CB - 1) / IN_CB;
  const int nchunk = in_nchunk(P);
  hipLaunchKernelGGL((instnorm_partial_kernel<T, 0>),
                     dim3(N * ncb * nchunk), dim3(IN_THREADS), 0, stream, x,
                     (const T*)nullptr, (const float*)nullptr, partials, N, C,
                     P, nchunk);
  hipLaunchKernelGGL(instnorm_finalize_fwd_kernel, dim3(N * ncb), dim3(64), 0,
                     stream, partials, mean, rstd, N, C, P, nchunk, eps);
  const long total = (long)N * P * C;
  long blocks = (total + IN_THREADS - 1) / IN_THREADS;
  if (blocks > 16384) blocks = 16384;
  hipLaunchKernelGGL((instnorm_apply_fwd_kernel<T>), dim3((int)blocks),
                     dim3(IN_THREADS), 0, stream, x, y, mean, rstd, total, C,
                     P);
}

template <typename T>
static void instnorm_bwd_t(const T* x, const T* dy, const float* mean,
                           const float* rstd, float* gmean, float* gxmean,
                           float* partials, T* dx, int N, int C, long P,
                           hipStream_t stream) {
  const int ncb = (C + IN_CB - 1) / IN_CB;
  const int nchunk = in_nchunk(P);
  hipLaunchKernelGGL((instnorm_partial_kernel<T, 1>),
                     dim3(N * ncb * nchunk), dim3(IN_THREADS), 0, stream, x,
                     dy, mean, partials, N, C, P, nchunk);
  hipLaunchKernelGGL(instnorm_finalize_bwd_kernel, dim3(N * ncb), dim3(64), 0,
                     stream, partials, gmean, gxmean, N, C, P, nchunk);
  const long total = (long)N * P * C;
  long blocks = (total + IN_THREADS - 1) / IN_THREADS;
  if (blocks > 16384) blocks = 16384;
  hipLaunchKernelGGL((instnorm_apply_bwd_kernel<T>), dim3((int)blocks),
                     dim3(IN_THREADS), 0, stream, x, dy, dx, mean, rstd,
                     gmean, gxmean, total, C, P);
}

void flowhip_instnorm_cl_fwd_launch(const void* x, void* y, float* mean,
                                    float* rstd, float* partials, int N,
                                    int C, long P, float eps, int is_bf16,
                                    hipStream_t stream) {
  if (is_bf16)
    instnorm_fwd_t((const __hip_bfloat16*)x, (__hip_bfloat16*)y, mean, rstd,
                   partials, N, C, P, eps, stream);
  else
    instnorm_fwd_t((const float*)x, (float*)y, mean, rstd, partials, N, C, P,
                   eps, stream);
}

void flowhip_instnorm_cl_bwd_launch(const void* x, const void* dy,
                                    const float* mean, const float* rstd,
                                    float* gmean, float* gxmean,
                                    float* partials, void* dx, int N, int C,
                                    long P, int is_bf16, hipStream_t stream) {
  if (is_bf16)
    instnorm_bwd_t((const __hip_bfloat16*)x, (const __hip_bfloat16*)dy, mean,
                   rstd, gmean, gxmean, partials, (__hip_bfloat16*)dx, N, C,
                   P, stream);
  else
    instnorm_bwd_t((const float*)x, (const float*)dy, mean, rstd, gmean,
                   gxmean, partials, (float*)dx, N, C, P, stream);
}
