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
// Backward: fused HIP kernels below (bwd-data gather + register-accumulated
// weight gradient); Python (functional_nconv) only precomputes the
// elementwise dnomin/ddenom terms and reconstructs denom from cout.

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

// ---------------------------------------------------------------------------
// Fused backward (replaces the MIOpen-composite path, which fell into
// naive_conv wrw/bwd fallback kernels for these tiny-channel full-res
// shapes — see profiles/).
//
// Given dnomin = dout/(denom+eps) and ddenom (precomputed elementwise):
//   bwd_data:  g = convT(dnomin, w); gd = convT(ddenom, w)
//              ddata = conf * g ;  dconf = data * g + gd
//   wrw:       dw[co,ci,d] = sum_p dnomin[co,p] (data*conf)[ci,p+d]
//                          + ddenom[co,p] conf[ci,p+d]
// convT at stride 1, same padding: out[p] = sum_d in[p - d + pad] w[d].
// ---------------------------------------------------------------------------

template <int K>
__global__ __launch_bounds__(NC_THREADS) void nconv_bwd_data_kernel(
    const float* __restrict__ dnomin,  // (N, Co, H, W)
    const float* __restrict__ ddenom,  // (N, Co, H, W)
    const float* __restrict__ data,    // (N, Ci, H, W)
    const float* __restrict__ conf,    // (N, Ci, H, W)
    const float* __restrict__ weight,  // (Co, Ci, K, K)
    float* __restrict__ ddata,         // (N, Ci, H, W)
    float* __restrict__ dconf,         // (N, Ci, H, W)
    int N, int Ci, int Co, int H, int W) {
  __shared__ float wsh[NC_MAXW];
  const int nw = Co * Ci * K * K;
  for (int i = threadIdx.x; i < nw; i += NC_THREADS) wsh[i] = weight[i];
  __syncthreads();

  const long total = (long)N * Ci * H * W;
  for (long idx = (long)blockIdx.x * NC_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * NC_THREADS) {
    long t = idx;
    const int x = t % W; t /= W;
    const int y = t % H; t /= H;
    const int ci = t % Ci; t /= Ci;
    const int n = t;
    const long plane = (long)H * W;

    float g = 0.f, gd = 0.f;
#pragma unroll
    for (int ky = 0; ky < K; ++ky) {
      const int yy = y - ky + K / 2;  // transposed-conv gather position
      if (yy < 0 || yy >= H) continue;
#pragma unroll
      for (int kx = 0; kx < K; ++kx) {
        const int xx = x - kx + K / 2;
        if (xx < 0 || xx >= W) continue;
        for (int co = 0; co < Co; ++co) {
          const float w = wsh[((co * Ci + ci) * K + ky) * K + kx];
          const long off = ((long)n * Co + co) * plane + (long)yy * W + xx;
          g += w * dnomin[off];
          gd += w * ddenom[off];
        }
      }
    }
    const long p = ((long)n * Ci + ci) * plane + (long)y * W + x;
    const float c = conf[p];
    const float d = data[p];
    ddata[p] = c * g;
    dconf[p] = d * g + gd;
  }
}

// Weight gradient: grid-stride pixel loop; each thread accumulates the
// FULL (Ci*K*K) slice for ONE co in registers, wave-reduces, block leader
// atomically adds into global dweight.
template <int K, int CI>
__global__ __launch_bounds__(NC_THREADS) void nconv_wrw_kernel(
    const float* __restrict__ dnomin,  // (N, Co, H, W)
    const float* __restrict__ ddenom,  // (N, Co, H, W)
    const float* __restrict__ data,    // (N, Ci, H, W)
    const float* __restrict__ conf,    // (N, Ci, H, W)
    float* __restrict__ dweight,       // (Co, Ci, K, K), zero-init
    int N, int Co, int H, int W, long px_per_block) {
  const int co = blockIdx.y;
  const long plane = (long)H * W;
  const long total = (long)N * plane;

  float acc[CI * K * K];
#pragma unroll
  for (int i = 0; i < CI * K * K; ++i) acc[i] = 0.f;

  const long start = (long)blockIdx.x * px_per_block + threadIdx.x;
  const long end = min(total, (long)(blockIdx.x + 1) * px_per_block);
  for (long idx = start; idx < end; idx += NC_THREADS) {
    long t = idx;
    const int x = t % W; t /= W;
    const int y = t % H; t /= H;
    const int n = t;

    const float gn = dnomin[((long)n * Co + co) * plane + (long)y * W + x];
    const float gd = ddenom[((long)n * Co + co) * plane + (long)y * W + x];

#pragma unroll
    for (int ci = 0; ci < CI; ++ci) {
      const float* dch = data + ((long)n * CI + ci) * plane;
      const float* cch = conf + ((long)n * CI + ci) * plane;
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
          acc[(ci * K + ky) * K + kx] += gn * dc + gd * c;
        }
      }
    }
  }

  // wave shuffle-reduce each accumulator, then one atomic per wave per value
  const int lane = threadIdx.x & 63;
#pragma unroll
  for (int i = 0; i < CI * K * K; ++i) {
    float v = acc[i];
#pragma unroll
    for (int s = 32; s > 0; s >>= 1) v += __shfl_down(v, s, 64);
    if (lane == 0 && v != 0.f)
      atomicAdd(&dweight[(long)co * CI * K * K + i], v);
  }
}

void flowhip_nconv_bwd_data_launch(const float* dnomin, const float* ddenom,
                                   const float* data, const float* conf,
                                   const float* weight, float* ddata,
                                   float* dconf, int N, int Ci, int Co, int H,
                                   int W, int K, hipStream_t stream) {
  const long total = (long)N * Ci * H * W;
  int blocks = (int)((total + NC_THREADS - 1) / NC_THREADS);
  if (blocks > 16384) blocks = 16384;
  dim3 grid(blocks), block(NC_THREADS);
  switch (K) {
    case 1: hipLaunchKernelGGL((nconv_bwd_data_kernel<1>), grid, block, 0, stream, dnomin, ddenom, data, conf, weight, ddata, dconf, N, Ci, Co, H, W); break;
    case 3: hipLaunchKernelGGL((nconv_bwd_data_kernel<3>), grid, block, 0, stream, dnomin, ddenom, data, conf, weight, ddata, dconf, N, Ci, Co, H, W); break;
    case 5: hipLaunchKernelGGL((nconv_bwd_data_kernel<5>), grid, block, 0, stream, dnomin, ddenom, data, conf, weight, ddata, dconf, N, Ci, Co, H, W); break;
    default: abort();
  }
}

template <int K>
static void wrw_ci(const float* dnomin, const float* ddenom,
                   const float* data, const float* conf, float* dweight,
                   int N, int Ci, int Co, int H, int W, hipStream_t stream) {
  const long total = (long)N * H * W;
  // enough blocks to fill the chip, few enough to keep atomics cheap
  int blocks = 1024;
  long px_per_block = (total + blocks - 1) / blocks;
  if (px_per_block < NC_THREADS) {
    px_per_block = NC_THREADS;
    blocks = (int)((total + px_per_block - 1) / px_per_block);
  }
  dim3 grid(blocks, Co), block(NC_THREADS);
  switch (Ci) {
    case 1: hipLaunchKernelGGL((nconv_wrw_kernel<K, 1>), grid, block, 0, stream, dnomin, ddenom, data, conf, dweight, N, Co, H, W, px_per_block); break;
    case 2: hipLaunchKernelGGL((nconv_wrw_kernel<K, 2>), grid, block, 0, stream, dnomin, ddenom, data, conf, dweight, N, Co, H, W, px_per_block); break;
    case 4: hipLaunchKernelGGL((nconv_wrw_kernel<K, 4>), grid, block, 0, stream, dnomin, ddenom, data, conf, dweight, N, Co, H, W, px_per_block); break;
    default: abort();
  }
}

void flowhip_nconv_wrw_launch(const float* dnomin, const float* ddenom,
                              const float* data, const float* conf,
                              float* dweight, int N, int Ci, int Co, int H,
                              int W, int K, hipStream_t stream) {
  switch (K) {
    case 1: wrw_ci<1>(dnomin, ddenom, data, conf, dweight, N, Ci, Co, H, W, stream); break;
    case 3: wrw_ci<3>(dnomin, ddenom, data, conf, dweight, N, Ci, Co, H, W, stream); break;
    case 5: wrw_ci<5>(dnomin, ddenom, data, conf, dweight, N, Ci, Co, H, W, stream); break;
    default: abort();
  }
}
