// Fused normalized-convolution forward (kernel #6 of SURVEY.md §2.2).
//
//   denom = conv(conf, w);  nomin = conv(data*conf, w)
//   nconv = nomin / (denom + 1e-20) [+ bias]
//   cout  = denom / sum_per_outchannel(w)
// (reference nconv_modules.py:164-199; w >= 0 via softplus, applied by the
// caller). Stride 1, same padding (K//2), zero pad, groups=1, K in {1,3,5},
// Cin/Cout <= 8 — the NCUP configuration space.
//
// The reference issues two full conv2d calls plus elementwise div/mul; this
// kernel reads (data, conf) once and writes (nconv, cout) once — the op is
// memory-bound at full image resolution. One thread per output pixel
// computes every output channel; the (tiny) weight tensor and its
// per-channel sums live in LDS.
//
// Backward runs as a torch/MIOpen composition in Python (functional_nconv),
// using the saved cout to reconstruct denom.

#include "common.h"

#define NC_THREADS 256
#define NC_MAXW (8 * 8 * 25)

template <int K>
__global__ __launch_bounds__(NC_THREADS) void nconv_fwd_kernel(
    const float* __restrict__ data,   // (N, Ci, H, W)
    const float* __restrict__ conf,   // (N, Ci, H, W)
    const float* __restrict__ weight, // (Co, Ci, K, K)
    const float* __restrict__ bias,   // (Co) or nullptr
    float* __restrict__ out,          // (N, Co, H, W)
    float* __restrict__ cout,         // (N, Co, H, W)
    int N, int Ci, int Co, int H, int W) {
  __shared__ float wsh[NC_MAXW];
  __shared__ float wsum[8];

  const int nw = Co * Ci * K * K;
  for (int i = threadIdx.x; i < nw; i += NC_THREADS) wsh[i] = weight[i];
  __syncthreads();
  if (threadIdx.x < Co) {
    float s = 0.f;
    for (int i = 0; i < Ci * K * K; ++i) s += wsh[threadIdx.x * Ci * K * K + i];
    wsum[threadIdx.x] = 1.0f / s;
  }
  __syncthreads();

  const long total = (long)N * H * W;
  for (long idx = (long)blockIdx.x * NC_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * NC_THREADS) {
    long t = idx;
    const int x = t % W; t /= W;
    const int y = t % H; t /= H;
    const int n = t;
    const long plane = (long)H * W;

    float denom[8], nomin[8];
#pragma unroll
    for (int co = 0; co < 8; ++co) { denom[co] = 0.f; nomin[co] = 0.f; }

    for (int ci = 0; ci < Ci; ++ci) {
      const float* dch = data + ((long)n * Ci + ci) * plane;
      const float* cch = conf + ((long)n * Ci + ci) * plane;
#pragma unroll
      for (int ky = 0; ky < K; ++ky) {
        const int yy = y + ky - K / 2;
        if (yy < 0 || yy >= H) continue;
#pragma unroll
        for (int kx = 0; kx < K; ++kx) {
          const int xx = x + kx - K / 2;
          if (xx < 0 || xx >= W) continue;
          const float c = cch[(long)yy * W + xx];
          const float dc = dch[(long)yy * W + xx] * c;
          for (int co = 0; co < Co; ++co) {
            const float w = wsh[((co * Ci + ci) * K + ky) * K + kx];
            denom[co] += w * c;
            nomin[co] += w * dc;
          }
        }
      }
    }

    for (int co = 0; co < Co; ++co) {
      float v = nomin[co] / (denom[co] + 1e-20f);
      if (bias != nullptr) v += bias[co];
      out[((long)n * Co + co) * plane + (long)y * W + x] = v;
      cout[((long)n * Co + co) * plane + (long)y * W + x] =
          denom[co] * wsum[co];
    }
  }
}

void flowhip_nconv_fwd_launch(const float* data, const float* conf,
                              const float* weight, const float* bias,
                              float* out, float* cout, int N, int Ci, int Co,
                              int H, int W, int K, hipStream_t stream) {
  const long total = (long)N * H * W;
  int blocks = (int)((total + NC_THREADS - 1) / NC_THREADS);
  if (blocks > 16384) blocks = 16384;
  dim3 grid(blocks), block(NC_THREADS);
  switch (K) {
    case 1:
      hipLaunchKernelGGL((nconv_fwd_kernel<1>), grid, block, 0, stream, data,
                         conf, weight, bias, out, cout, N, Ci, Co, H, W);
      break;
    case 3:
      hipLaunchKernelGGL((nconv_fwd_kernel<3>), grid, block, 0, stream, data,
                         conf, weight, bias, out, cout, N, Ci, Co, H, W);
      break;
    case 5:
      hipLaunchKernelGGL((nconv_fwd_kernel<5>), grid, block, 0, stream, data,
                         conf, weight, bias, out, cout, N, Ci, Co, H, W);
      break;
    default:
      abort();
  }
}
