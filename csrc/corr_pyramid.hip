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
// Templated over the element type: fp32 (reference parity) or bf16 (the
// HBM-resident bf16 pyramid of the north star / BASELINE config 5);
// arithmetic is fp32 either way. LDS is allocated dynamically from the
// actual map size — level-0 maps up to ~31k fp32 / ~62k bf16 cells fit the
// 160 KiB budget (KITTI-submission 47x156 and every BASELINE shape
// included; round-1's static 56x128 cap silently dropped KITTI to the torch
// chain — VERDICT r01).
//
// Block = one (b,i) map; 256 threads cooperate via the LDS staging buffer.

#include "common.h"

#define PYR_THREADS 256
#define PYR_LDS_BUDGET (160 * 1024)

static inline long pyr_lds_bytes(int H0, int W0, int esz) {
  const long s0 = ((long)H0 * W0 * esz + 15) / 16 * 16;
  const long s1 = ((long)(H0 / 2) * (W0 / 2) * esz + 15) / 16 * 16;
  const long s2 = ((long)(H0 / 4) * (W0 / 4) * esz + 15) / 16 * 16;
  return s0 + s1 + s2;
}

template <typename scalar_t>
__global__ __launch_bounds__(PYR_THREADS) void corr_pyramid_fwd_kernel(
    const scalar_t* __restrict__ corr,  // (BP, H0, W0)
    scalar_t* __restrict__ l1, scalar_t* __restrict__ l2,
    scalar_t* __restrict__ l3, int BP, int H0, int W0, int nlev) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  const int H1 = H0 / 2, W1 = W0 / 2;
  const int H2 = H1 / 2, W2 = W1 / 2;
  const int H3 = H2 / 2, W3 = W2 / 2;

  scalar_t* s0 = (scalar_t*)smem;
  scalar_t* s1 = (scalar_t*)(smem + (((long)H0 * W0 * sizeof(scalar_t) + 15) /
                                     16 * 16));
  scalar_t* s2 = (scalar_t*)((char*)s1 + (((long)H1 * W1 * sizeof(scalar_t) +
                                           15) / 16 * 16));

  const int bp = blockIdx.x;
  const scalar_t* map = corr + (long)bp * H0 * W0;
  for (int i = threadIdx.x; i < H0 * W0; i += PYR_THREADS) s0[i] = map[i];
  __syncthreads();

  scalar_t* o1 = l1 + (long)bp * H1 * W1;
  for (int i = threadIdx.x; i < H1 * W1; i += PYR_THREADS) {
    const int y = i / W1, x = i - y * W1;
    const float v = 0.25f * ((float)s0[(2 * y) * W0 + 2 * x] +
                             (float)s0[(2 * y) * W0 + 2 * x + 1] +
                             (float)s0[(2 * y + 1) * W0 + 2 * x] +
                             (float)s0[(2 * y + 1) * W0 + 2 * x + 1]);
    s1[i] = (scalar_t)v;
    o1[i] = (scalar_t)v;
  }
  if (nlev < 3) return;
  __syncthreads();

  scalar_t* o2 = l2 + (long)bp * H2 * W2;
  for (int i = threadIdx.x; i < H2 * W2; i += PYR_THREADS) {
    const int y = i / W2, x = i - y * W2;
    const float v = 0.25f * ((float)s1[(2 * y) * W1 + 2 * x] +
                             (float)s1[(2 * y) * W1 + 2 * x + 1] +
                             (float)s1[(2 * y + 1) * W1 + 2 * x] +
                             (float)s1[(2 * y + 1) * W1 + 2 * x + 1]);
    s2[i] = (scalar_t)v;
    o2[i] = (scalar_t)v;
  }
  if (nlev < 4) return;
  __syncthreads();

  scalar_t* o3 = l3 + (long)bp * H3 * W3;
  for (int i = threadIdx.x; i < H3 * W3; i += PYR_THREADS) {
    const int y = i / W3, x = i - y * W3;
    o3[i] = (scalar_t)(0.25f * ((float)s2[(2 * y) * W2 + 2 * x] +
                                (float)s2[(2 * y) * W2 + 2 * x + 1] +
                                (float)s2[(2 * y + 1) * W2 + 2 * x] +
                                (float)s2[(2 * y + 1) * W2 + 2 * x + 1]));
  }
}

template <typename scalar_t>
__global__ __launch_bounds__(PYR_THREADS) void corr_pyramid_bwd_kernel(
    const scalar_t* __restrict__ g0,  // (BP, H0, W0) or nullptr
    const scalar_t* __restrict__ g1,  // (BP, H1, W1) or nullptr
    const scalar_t* __restrict__ g2,
    const scalar_t* __restrict__ g3,
    scalar_t* __restrict__ dcorr,     // (BP, H0, W0)
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

    float v = g0 ? (float)g0[idx] : 0.0f;
    if (g1 && (y >> 1) < H1 && (x >> 1) < W1)
      v += 0.25f * (float)g1[((long)bp * H1 + (y >> 1)) * W1 + (x >> 1)];
    if (g2 && (y >> 2) < H2 && (x >> 2) < W2)
      v += 0.0625f * (float)g2[((long)bp * H2 + (y >> 2)) * W2 + (x >> 2)];
    if (g3 && (y >> 3) < H3 && (x >> 3) < W3)
      v += 0.015625f * (float)g3[((long)bp * H3 + (y >> 3)) * W3 + (x >> 3)];
    dcorr[idx] = (scalar_t)v;
  }
}

bool flowhip_corr_pyramid_fwd_launch(const void* corr, void* l1, void* l2,
                                     void* l3, int BP, int H0, int W0,
                                     int nlev, int is_bf16,
                                     hipStream_t stream) {
  if (nlev < 2 || nlev > 4) return false;
  const long lds = pyr_lds_bytes(H0, W0, is_bf16 ? 2 : 4);
  if (lds > PYR_LDS_BUDGET) return false;
  if (is_bf16)
    hipLaunchKernelGGL(corr_pyramid_fwd_kernel<__bf16>, dim3(BP),
                       dim3(PYR_THREADS), (size_t)lds, stream,
                       (const __bf16*)corr, (__bf16*)l1, (__bf16*)l2,
                       (__bf16*)l3, BP, H0, W0, nlev);
  else
    hipLaunchKernelGGL(corr_pyramid_fwd_kernel<float>, dim3(BP),
                       dim3(PYR_THREADS), (size_t)lds, stream,
                       (const float*)corr, (float*)l1, (float*)l2, (float*)l3,
                       BP, H0, W0, nlev);
  return true;
}

bool flowhip_corr_pyramid_fits(int H0, int W0, int is_bf16) {
  return pyr_lds_bytes(H0, W0, is_bf16 ? 2 : 4) <= PYR_LDS_BUDGET;
}

void flowhip_corr_pyramid_bwd_launch(const void* g0, const void* g1,
                                     const void* g2, const void* g3,
                                     void* dcorr, long total, int H0, int W0,
                                     int is_bf16, hipStream_t stream) {
  long blocks = (total + PYR_THREADS - 1) / PYR_THREADS;
  if (blocks > 32768) blocks = 32768;
  if (is_bf16)
    hipLaunchKernelGGL(corr_pyramid_bwd_kernel<__bf16>, dim3((int)blocks),
                       dim3(PYR_THREADS), 0, stream, (const __bf16*)g0,
                       (const __bf16*)g1, (const __bf16*)g2,
                       (const __bf16*)g3, (__bf16*)dcorr, total, H0, W0);
  else
    hipLaunchKernelGGL(corr_pyramid_bwd_kernel<float>, dim3((int)blocks),
                       dim3(PYR_THREADS), 0, stream, (const float*)g0,
                       (const float*)g1, (const float*)g2, (const float*)g3,
                       (float*)dcorr, total, H0, W0);
}
