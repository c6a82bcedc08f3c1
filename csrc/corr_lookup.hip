// Fused correlation-pyramid window lookup (kernel #3 of SURVEY.md §2.2) —
// the per-iteration hot op of the RAFT loop (reference core/corr.py:23-44 +
// utils/utils.py:59-67 grid_sample).
//
// For each target pixel i of image 1 (which owns its private (Hl, Wl)
// correlation map at every pyramid level), sample a (2r+1)^2 bilinear window
// around coords[i]/2^l and write it channel-major:
//     out[b, l*K2 + a*K + c, i]   samples   (x + (a-r), y + (c-r))
// — note the x-offset-MAJOR channel order: the reference builds delta as
// meshgrid(dy, dx) but adds it to (x, y)-ordered coords (corr.py:31-37), so
// the first window index offsets x. The symmetric window makes coverage
// identical; the channel layout must match for weight compatibility.
//
// Sampling semantics = grid_sample(align_corners=True, padding_mode=zeros):
// out-of-range corner pixels contribute zero.
//
// Parallelization: thread = one target pixel, block = 256 consecutive
// pixels, gridDim.y = pyramid level. Each thread loops over its level's
// 81 taps: 4 corner loads from its private window (L1/L2-resident, ~25
// lines) + one fully coalesced store per tap (adjacent lanes = adjacent
// pixels = adjacent addresses in the (B, C, P) output). The window offsets
// are compile-time unrolled — no host->device delta transfer per iteration
// (fixes SURVEY.md §2.9 quirk 8) and the kernel is hipGraph-safe.
//
// Backward: grads flow to the pyramid levels only (coords are detached
// every iteration in RAFT — raft.py:122). Each thread owns its pixel's private
// map, so the 4-corner scatter-adds need no atomics.

#include "common.h"

#define LK_THREADS 256

template <int R, typename scalar_t>
__global__ __launch_bounds__(LK_THREADS) void corr_lookup_fwd_kernel(
    const scalar_t* __restrict__ level,  // (B*P, Hl, Wl)
    const float* __restrict__ coords,    // (B, 2, H, W)
    float* __restrict__ out,             // (B, L*K2, H, W)
    int BP, int P, int Hl, int Wl, int l, int L) {
  constexpr int K = 2 * R + 1;
  constexpr int K2 = K * K;

  const int pix = blockIdx.x * LK_THREADS + threadIdx.x;
  if (pix >= BP) return;
  const int b = pix / P;
  const int i = pix % P;

  const float inv = 1.0f / (float)(1 << l);
  const float cx = coords[((long)b * 2 + 0) * P + i] * inv;
  const float cy = coords[((long)b * 2 + 1) * P + i] * inv;

  const scalar_t* map = level + (long)pix * Hl * Wl;
  float* outb = out + ((long)b * L * K2 + (long)l * K2) * P + i;

#pragma unroll
  for (int a = 0; a < K; ++a) {    // x-offset index (major)
    const float sx = cx + (a - R);
    const float fx0 = floorf(sx);
    const int x0 = (int)fx0;
    const float wx1 = sx - fx0;
    const float wx0 = 1.0f - wx1;
    const bool vx0 = (x0 >= 0) & (x0 < Wl);
    const bool vx1 = (x0 + 1 >= 0) & (x0 + 1 < Wl);
#pragma unroll
    for (int c = 0; c < K; ++c) {  // y-offset index
      const float sy = cy + (c - R);
      const float fy0 = floorf(sy);
      const int y0 = (int)fy0;
      const float wy1 = sy - fy0;
      const float wy0 = 1.0f - wy1;
      const bool vy0 = (y0 >= 0) & (y0 < Hl);
      const bool vy1 = (y0 + 1 >= 0) & (y0 + 1 < Hl);

      float v = 0.0f;
      if (vx0 & vy0) v += wx0 * wy0 * (float)map[(long)y0 * Wl + x0];
      if (vx1 & vy0) v += wx1 * wy0 * (float)map[(long)y0 * Wl + x0 + 1];
      if (vx0 & vy1) v += wx0 * wy1 * (float)map[(long)(y0 + 1) * Wl + x0];
      if (vx1 & vy1) v += wx1 * wy1 * (float)map[(long)(y0 + 1) * Wl + x0 + 1];

      outb[(long)(a * K + c) * P] = v;
    }
  }
}

template <int R, typename scalar_t>
__global__ __launch_bounds__(LK_THREADS) void corr_lookup_bwd_kernel(
    const float* __restrict__ gout,    // (B, L*K2, H, W)
    const float* __restrict__ coords,  // (B, 2, H, W)
    scalar_t* __restrict__ glevel,     // (B*P, Hl, Wl), zero-initialized
    int BP, int P, int Hl, int Wl, int l, int L) {
  constexpr int K = 2 * R + 1;
  constexpr int K2 = K * K;

  const int pix = blockIdx.x * LK_THREADS + threadIdx.x;
  if (pix >= BP) return;
  const int b = pix / P;
  const int i = pix % P;

  const float inv = 1.0f / (float)(1 << l);
  const float cx = coords[((long)b * 2 + 0) * P + i] * inv;
  const float cy = coords[((long)b * 2 + 1) * P + i] * inv;

  scalar_t* gmap = glevel + (long)pix * Hl * Wl;
  const float* gin = gout + ((long)b * L * K2 + (long)l * K2) * P + i;

#pragma unroll
  for (int a = 0; a < K; ++a) {
    const float sx = cx + (a - R);
    const float fx0 = floorf(sx);
    const int x0 = (int)fx0;
    const float wx1 = sx - fx0;
    const float wx0 = 1.0f - wx1;
    const bool vx0 = (x0 >= 0) & (x0 < Wl);
    const bool vx1 = (x0 + 1 >= 0) & (x0 + 1 < Wl);
#pragma unroll
    for (int c = 0; c < K; ++c) {
      const float sy = cy + (c - R);
      const float fy0 = floorf(sy);
      const int y0 = (int)fy0;
      const float wy1 = sy - fy0;
      const float wy0 = 1.0f - wy1;
      const bool vy0 = (y0 >= 0) & (y0 < Hl);
      const bool vy1 = (y0 + 1 >= 0) & (y0 + 1 < Hl);

      const float g = gin[(long)(a * K + c) * P];
      // exclusive ownership of the pixel's map: plain read-modify-write
      if (vx0 & vy0) {
        scalar_t* p = gmap + (long)y0 * Wl + x0;
        *p = (scalar_t)((float)*p + wx0 * wy0 * g);
      }
      if (vx1 & vy0) {
        scalar_t* p = gmap + (long)y0 * Wl + x0 + 1;
        *p = (scalar_t)((float)*p + wx1 * wy0 * g);
      }
      if (vx0 & vy1) {
        scalar_t* p = gmap + (long)(y0 + 1) * Wl + x0;
        *p = (scalar_t)((float)*p + wx0 * wy1 * g);
      }
      if (vx1 & vy1) {
        scalar_t* p = gmap + (long)(y0 + 1) * Wl + x0 + 1;
        *p = (scalar_t)((float)*p + wx1 * wy1 * g);
      }
    }
  }
}

template <int R>
static void lookup_fwd_level(const float* level, const float* coords,
                             float* out, int BP, int P, int Hl, int Wl, int l,
                             int L, hipStream_t stream) {
  dim3 grid(fh_cdiv(BP, LK_THREADS));
  hipLaunchKernelGGL((corr_lookup_fwd_kernel<R, float>), grid,
                     dim3(LK_THREADS), 0, stream, level, coords, out, BP, P,
                     Hl, Wl, l, L);
}

template <int R>
static void lookup_bwd_level(const float* gout, const float* coords,
                             float* glevel, int BP, int P, int Hl, int Wl,
                             int l, int L, hipStream_t stream) {
  dim3 grid(fh_cdiv(BP, LK_THREADS));
  hipLaunchKernelGGL((corr_lookup_bwd_kernel<R, float>), grid,
                     dim3(LK_THREADS), 0, stream, gout, coords, glevel, BP, P,
                     Hl, Wl, l, L);
}

void flowhip_corr_lookup_fwd_launch(const float* level, const float* coords,
                                    float* out, int BP, int P, int Hl, int Wl,
                                    int l, int L, int radius,
                                    hipStream_t stream) {
  switch (radius) {
    case 3: lookup_fwd_level<3>(level, coords, out, BP, P, Hl, Wl, l, L, stream); break;
    case 4: lookup_fwd_level<4>(level, coords, out, BP, P, Hl, Wl, l, L, stream); break;
    default: abort();
  }
}

void flowhip_corr_lookup_bwd_launch(const float* gout, const float* coords,
                                    float* glevel, int BP, int P, int Hl,
                                    int Wl, int l, int L, int radius,
                                    hipStream_t stream) {
  switch (radius) {
    case 3: lookup_bwd_level<3>(gout, coords, glevel, BP, P, Hl, Wl, l, L, stream); break;
    case 4: lookup_bwd_level<4>(gout, coords, glevel, BP, P, Hl, Wl, l, L, stream); break;
    default: abort();
  }
}
