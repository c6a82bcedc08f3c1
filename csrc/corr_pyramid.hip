// Fused correlation-pyramid build (kernel #2 of SURVEY.md §2.2; reference
// core/corr.py:19-21 — three chained F.avg_pool2d(2,2) calls).
//
// Forward: one kernel reads each (Hl0, Wl0) map once through LDS and emits
// all downsampled levels (the torch chain re-reads every intermediate level
// from HBM and runs 3 kernels + allocator traffic; this is one pass at the
// read-once lower bound). Level 0 is the input itself (aliased on the
// Python side).
//
// Backward: dcorr[y,x] = g0[y,x] + g1[y/2,x/2]/4 + g2[y/4,x/4]/16 +
// g3[y/8,x/8]/64 — one gather kernel replacing the 3-deep
// avg_pool2d-backward chain + intermediate adds. Cells in rows/cols dropped
// by floor-division pooling receive no higher-level contribution, matching
// avg_pool2d(2,2) exactly.
//
// Block = one (b,i) map; 256 threads cooperate via an LDS staging buffer
// (cap 56x128 fp32 = 28 KB; larger maps fall back to the torch chain).

#include "common.h"

#define PYR_THREADS 256
#define PYR_MAX_MAP (56 * 128)
#define PYR_MAX_L1 (28 * 64)
#define PYR_MAX_L2 (14 * 32)

__global__ __launch_bounds__(PYR_THREADS) void corr_pyramid_fwd_kernel(
    const float* __restrict__ corr,  // (BP, H0, W0)
    float* __restrict__ l1, float* __restrict__ l2, float* __restrict__ l3,
    int BP, int H0, int W0, int nlev) {
  __shared__ float s0[PYR_MAX_MAP];
  __shared__ float s1[PYR_MAX_L1];
  __shared__ float s2[PYR_MAX_L2];

  const int bp = blockIdx.x;
  const int H1 = H0 / 2, W1 = W0 / 2;
  const int H2 = H1 / 2, W2 = W1 / 2;
  const int H3 = H2 / 2, W3 = W2 / 2;

  const float* map = corr + (long)bp * H0 * W0;
  for (int i = threadIdx.x; i < H0 * W0; i += PYR_THREADS) s0[i] = map[i];
  __syncthreads();

  float* o1 = l1 + (long)bp * H1 * W1;
  for (int i = threadIdx.x; i < H1 * W1; i += PYR_THREADS) {
    const int y = i / W1, x = i - y * W1;
    const float v = 0.25f * (s0[(2 * y) * W0 + 2 * x] +
                             s0[(2 * y) * W0 + 2 * x + 1] +
                             s0[(2 * y + 1) * W0 + 2 * x] +
                             s0[(2 * y + 1) * W0 + 2 * x + 1]);
    s1[i] = v;
    o1[i] = v;
  }
  if (nlev < 3) return;
  __syncthreads();

  float* o2 = l2 + (long)bp * H2 * W2;
  for (int i = threadIdx.x; i < H2 * W2; i += PYR_THREADS) {
    const int y = i / W2, x = i - y * W2;
    const float v = 0.25f * (s1[(2 * y) * W1 + 2 * x] +
                             s1[(2 * y) * W1 + 2 * x + 1] +
                             s1[(2 * y + 1) * W1 + 2 * x] +
                             s1[(2 * y + 1) * W1 + 2 * x + 1]);
    s2[i] = v;
    o2[i] = v;
  }
  if (nlev < 4) return;
  __syncthreads();

  float* o3 = l3 + (long)bp * H3 * W3;
  for (int i = threadIdx.x; i < H3 * W3; i += PYR_THREADS) {
    const int y = i / W3, x = i - y * W3;
    o3[i] = 0.25f * (s2[(2 * y) * W2 + 2 * x] +
                     s2[(2 * y) * W2 + 2 * x + 1] +
                     s2[(2 * y + 1) * W2 + 2 * x] +
                     s2[(2 * y + 1) * W2 + 2 * x + 1]);
  }
}

__global__ __launch_bounds__(PYR_THREADS) void corr_pyramid_bwd_kernel(
    const float* __restrict__ g0,  // (BP, H0, W0) or nullptr
    const float* __restrict__ g1,  // (BP, H1, W1) or nullptr
    const float* __restrict__ g2,
    const float* __restrict__ g3,
    float* __restrict__ dcorr,     // (BP, H0, W0)
    long total, int H0, int W0) {
  const int H1 = H0 / 2, W1 = W0 / 2;
  const int H2 = H1 / 2, W2 = W1 / 2;
  const int H3 = H2 / 2, W3 = W2 / 2;

  for (long idx = (long)blockIdx.x * PYR_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * PYR_THREADS) {
    long t = idx;
    const int x = t % W0; t /= W0;
    const int y = t % H0; t /= H0;
    const long bp = t;

    float v = g0 ? g0[idx] : 0.0f;
    if (g1 && (y >> 1) < H1 && (x >> 1) < W1)
      v += 0.25f * g1[((long)bp * H1 + (y >> 1)) * W1 + (x >> 1)];
    if (g2 && (y >> 2) < H2 && (x >> 2) < W2)
      v += 0.0625f * g2[((long)bp * H2 + (y >> 2)) * W2 + (x >> 2)];
    if (g3 && (y >> 3) < H3 && (x >> 3) < W3)
      v += 0.015625f * g3[((long)bp * H3 + (y >> 3)) * W3 + (x >> 3)];
    dcorr[idx] = v;
  }
}

bool flowhip_corr_pyramid_fwd_launch(const float* corr, float* l1, float* l2,
                                     float* l3, int BP, int H0, int W0,
                                     int nlev, hipStream_t stream) {
  if (H0 * W0 > PYR_MAX_MAP || nlev < 2 || nlev > 4) return false;
  hipLaunchKernelGGL(corr_pyramid_fwd_kernel, dim3(BP), dim3(PYR_THREADS), 0,
                     stream, corr, l1, l2, l3, BP, H0, W0, nlev);
  return true;
}

void flowhip_corr_pyramid_bwd_launch(const float* g0, const float* g1,
                                     const float* g2, const float* g3,
                                     float* dcorr, long total, int H0, int W0,
                                     hipStream_t stream) {
  long blocks = (total + PYR_THREADS - 1) / PYR_THREADS;
  if (blocks > 32768) blocks = 32768;
  hipLaunchKernelGGL(corr_pyramid_bwd_kernel, dim3((int)blocks),
                     dim3(PYR_THREADS), 0, stream, g0, g1, g2, g3, dcorr,
                     total, H0, W0);
}
