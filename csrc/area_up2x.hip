// Exact-2x 'area' interpolation (reference upsampler.py:150: the GRU-state
// guidance is resized H/8 -> H/4 every iteration). For an exact integer
// upscale, mode='area' degenerates to nearest duplication; its torch
// backward is an atomic adaptive-avg-pool scatter (~170 us/call). These
// kernels are a plain gather each way (backward: each input cell sums its
// 2x2 duplicated outputs — no atomics).

#include "common.h"

#define AU_THREADS 256

__global__ __launch_bounds__(AU_THREADS) void area_up2x_fwd_kernel(
    const float* __restrict__ in,  // (N*C, H, W)
    float* __restrict__ out,       // (N*C, 2H, 2W)
    long total, int H, int W) {
  const int OW = 2 * W;
  for (long idx = (long)blockIdx.x * AU_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * AU_THREADS) {
    long t = idx;
    const int x = t % OW; t /= OW;
    const int y = t % (2 * H); t /= (2 * H);
    out[idx] = in[(t * H + (y >> 1)) * W + (x >> 1)];
  }
}

__global__ __launch_bounds__(AU_THREADS) void area_up2x_bwd_kernel(
    const float* __restrict__ gout,  // (N*C, 2H, 2W)
    float* __restrict__ gin,         // (N*C, H, W)
    long total_in, int H, int W) {
  const int OW = 2 * W;
  for (long idx = (long)blockIdx.x * AU_THREADS + threadIdx.x;
       idx < total_in; idx += (long)gridDim.x * AU_THREADS) {
    long t = idx;
    const int x = t % W; t /= W;
    const int y = t % H; t /= H;
    const long base = (t * 2 * H + 2 * y) * OW + 2 * x;
    gin[idx] = gout[base] + gout[base + 1] + gout[base + OW] +
               gout[base + OW + 1];
  }
}

void flowhip_area_up2x_fwd_launch(const float* in, float* out, long total,
                                  int H, int W, hipStream_t stream) {
  long blocks = (total + AU_THREADS - 1) / AU_THREADS;
  if (blocks > 16384) blocks = 16384;
  hipLaunchKernelGGL(area_up2x_fwd_kernel, dim3((int)blocks),
                     dim3(AU_THREADS), 0, stream, in, out, total, H, W);
}

void flowhip_area_up2x_bwd_launch(const float* gout, float* gin,
                                  long total_in, int H, int W,
                                  hipStream_t stream) {
  long blocks = (total_in + AU_THREADS - 1) / AU_THREADS;
  if (blocks > 16384) blocks = 16384;
  hipLaunchKernelGGL(area_up2x_bwd_kernel, dim3((int)blocks),
                     dim3(AU_THREADS), 0, stream, gout, gin, total_in, H, W);
}

// Fused nearest-2x upsample + channel concat for the NConvUNet decoder
// skips (nconv_modules.py:128-134: F.interpolate(nearest) + cat, done for
// both data and conf every iteration):
//   out[:, :C1]  = low[n, c, y/2, x/2]   (exact 2x nearest)
//   out[:, C1:]  = skip[n, c-C1, y, x]
// Backward: dlow = 2x2 sum gather (one kernel); dskip = channel-slice view
// (zero-copy on the Python side).
__global__ __launch_bounds__(AU_THREADS) void up2x_cat_fwd_kernel(
    const float* __restrict__ low,   // (N, C1, H/2, W/2)
    const float* __restrict__ skip,  // (N, C2, H, W)
    float* __restrict__ out,         // (N, C1+C2, H, W)
    long total, int C1, int C2, int H, int W) {
  const int HL = H / 2, WL = W / 2;
  for (long idx = (long)blockIdx.x * AU_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * AU_THREADS) {
    long t = idx;
    const int x = t % W; t /= W;
    const int y = t % H; t /= H;
    const int c = t % (C1 + C2); t /= (C1 + C2);
    const long n = t;
    float v;
    if (c < C1) {
      int yl = y >> 1, xl = x >> 1;
      if (yl >= HL) yl = HL - 1;  // odd-size guard (nearest floor clamp)
      if (xl >= WL) xl = WL - 1;
      v = low[((n * C1 + c) * HL + yl) * WL + xl];
    } else {
      v = skip[((n * C2 + (c - C1)) * H + y) * W + x];
    }
    out[idx] = v;
  }
}

void flowhip_up2x_cat_fwd_launch(const float* low, const float* skip,
                                 float* out, long total, int C1, int C2,
                                 int H, int W, hipStream_t stream) {
  long blocks = (total + AU_THREADS - 1) / AU_THREADS;
  if (blocks > 16384) blocks = 16384;
  hipLaunchKernelGGL(up2x_cat_fwd_kernel, dim3((int)blocks),
                     dim3(AU_THREADS), 0, stream, low, skip, out, total, C1,
                     C2, H, W);
}
