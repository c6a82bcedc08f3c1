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
// SEPARABLE-WINDOW FORM: the window offsets are integers, so
// floor(cx + (a-R)) = floor(cx) + (a-R) and the fractional weights
// (wx0,wx1,wy0,wy1) are IDENTICAL for all K^2 taps. The 4*K^2 corner
// gathers therefore collapse to one (K+1)x(K+1) patch read at
// (floor(cx)-R, floor(cy)-R) with a separable 2-tap blend:
//    tmp[a][c]  = wy0*patch[a][c] + wy1*patch[a][c+1]
//    out[a][c]  = wx0*tmp[a][c]   + wx1*tmp[a+1][c]
// (100 loads + ~250 fma vs 324 loads of the naive form). The backward is
// the transpose: the K^2 incoming grads splat onto the same (K+1)^2 patch
// with one read-modify-write per patch element (100 RMWs vs 1296) — each
// thread owns its pixel's private map, so no atomics.
//
// Parallelization: thread = one target pixel, 256-thread blocks over B*P.
// cl=1 lays the output channels-last, (B,H,W,C): per-thread tap stores/loads
// are then unit-stride (the NHWC layout the MIOpen conv consumers run in).
// The offsets are compile-time; no host->device delta transfer per
// iteration (fixes SURVEY.md §2.9 quirk 8) and the kernel is hipGraph-safe.

#include "common.h"

#define LK_THREADS 256

// all pyramid levels in ONE launch (blockIdx.y = level): a per-level
// launch at level 0 is only BP/256 workgroups (84 at the flagship shape)
// -- a 3x underfill of the 256-CU chip; batching the levels quadruples
// the resident blocks and cuts 8 launches per iteration to 2.
struct LkLevels {
  const void* p[4];
  void* g[4];
  int H[4];
  int W[4];
};

template <int R, typename scalar_t>
__global__ __launch_bounds__(LK_THREADS) void corr_lookup_fwd_kernel(
    LkLevels lv,                         // levels (B*P, Hl, Wl)
    const float* __restrict__ coords,    // (B, 2, H, W)
    scalar_t* __restrict__ out,          // (B, L*K2, H, W) NCHW or NHWC
    int BP, int P, int L, int cl, int ldc) {
  constexpr int K = 2 * R + 1;
  constexpr int K2 = K * K;

  const int l = blockIdx.y;
  const scalar_t* __restrict__ level = (const scalar_t*)lv.p[l];
  const int Hl = lv.H[l], Wl = lv.W[l];
  const int pix = blockIdx.x * LK_THREADS + threadIdx.x;
  if (pix >= BP) return;
  const int b = pix / P;
  const int i = pix % P;

  const float inv = 1.0f / (float)(1 << l);
  const float cx = coords[((long)b * 2 + 0) * P + i] * inv;
  const float cy = coords[((long)b * 2 + 1) * P + i] * inv;

  const float fx = floorf(cx), fy = floorf(cy);
  const int x0 = (int)fx - R;  // patch origin
  const int y0 = (int)fy - R;
  const float wx1 = cx - fx, wx0 = 1.0f - wx1;
  const float wy1 = cy - fy, wy0 = 1.0f - wy1;

  const scalar_t* map = level + (long)pix * Hl * Wl;
  const long tap_stride = cl ? 1 : (long)P;
  scalar_t* outb = cl ? out + ((long)b * P + i) * ldc + (long)l * K2
                      : out + ((long)b * L * K2 + (long)l * K2) * P + i;
  if (cl && l == 0) {
    // zero the channel-pad tail once (ldc > L*K2: the emitted tensor is a
    // narrow view of an 8-channel-aligned allocation so the NHWC conv
    // consumer's 16-B staging reads stay in-row)
    scalar_t* rowb = out + ((long)b * P + i) * ldc;
    for (int c = L * K2; c < ldc; ++c) rowb[c] = (scalar_t)0.0f;
  }

  const bool interior = (x0 >= 0) & (y0 >= 0) & (x0 + K < Wl) &
                        (y0 + K < Hl);
  if (interior) {
    // Fast path: stream patch ROWS (K+1 contiguous loads each — one base
    // address per row instead of per-element strided address math), blend
    // x within the row, then blend y across consecutive rows.
    float rprev[K], rcur[K];
#pragma unroll
    for (int j = 0; j <= K; ++j) {
      const scalar_t* row = map + (long)(y0 + j) * Wl + x0;
      float rv[K + 1];
#pragma unroll
      for (int a = 0; a <= K; ++a) rv[a] = (float)row[a];
#pragma unroll
      for (int a = 0; a < K; ++a) rcur[a] = wx0 * rv[a] + wx1 * rv[a + 1];
      if (j > 0) {
#pragma unroll
        for (int a = 0; a < K; ++a)
          outb[(long)(a * K + (j - 1)) * tap_stride] =
              (scalar_t)(wy0 * rprev[a] + wy1 * rcur[a]);
      }
#pragma unroll
      for (int a = 0; a < K; ++a) rprev[a] = rcur[a];
    }
    return;
  }

  float tprev[K];   // wy-blended column a-1
  float tcur[K];
#pragma unroll
  for (int a = 0; a <= K; ++a) {  // patch columns (x direction)
    const int xx = x0 + a;
    const bool vx = (xx >= 0) & (xx < Wl);
    // load patch column, zero out-of-range
    float col[K + 1];
#pragma unroll
    for (int j = 0; j <= K; ++j) {
      const int yy = y0 + j;
      col[j] = (vx & (yy >= 0) & (yy < Hl))
                   ? (float)map[(long)yy * Wl + xx] : 0.0f;
    }
#pragma unroll
    for (int c = 0; c < K; ++c) tcur[c] = wy0 * col[c] + wy1 * col[c + 1];
    if (a > 0) {
#pragma unroll
      for (int c = 0; c < K; ++c)
        outb[(long)((a - 1) * K + c) * tap_stride] =
            (scalar_t)(wx0 * tprev[c] + wx1 * tcur[c]);
    }
#pragma unroll
    for (int c = 0; c < K; ++c) tprev[c] = tcur[c];
  }
}

template <int R, typename scalar_t>
__global__ __launch_bounds__(LK_THREADS) void corr_lookup_bwd_kernel(
    const scalar_t* __restrict__ gout,  // (B, L*K2, H, W)
    const float* __restrict__ coords,  // (B, 2, H, W)
    LkLevels lv,                       // glevels (B*P, Hl, Wl); zeroed
                                       // (acc=0) or accumulated into (acc=1:
                                       // the iteration-chained grad buffer)
    int BP, int P, int L, int cl, int acc) {
  constexpr int K = 2 * R + 1;
  constexpr int K2 = K * K;

  const int l = blockIdx.y;
  scalar_t* __restrict__ glevel = (scalar_t*)lv.g[l];
  const int Hl = lv.H[l], Wl = lv.W[l];
  const int pix = blockIdx.x * LK_THREADS + threadIdx.x;
  if (pix >= BP) return;
  const int b = pix / P;
  const int i = pix % P;

  const float inv = 1.0f / (float)(1 << l);
  const float cx = coords[((long)b * 2 + 0) * P + i] * inv;
  const float cy = coords[((long)b * 2 + 1) * P + i] * inv;

  const float fx = floorf(cx), fy = floorf(cy);
  const int x0 = (int)fx - R;
  const int y0 = (int)fy - R;
  const float wx1 = cx - fx, wx0 = 1.0f - wx1;
  const float wy1 = cy - fy, wy0 = 1.0f - wy1;

  scalar_t* gmap = glevel + (long)pix * Hl * Wl;
  const long tap_stride = cl ? 1 : (long)P;
  const scalar_t* gin = cl
      ? gout + ((long)b * P + i) * (L * K2) + (long)l * K2
      : gout + ((long)b * L * K2 + (long)l * K2) * P + i;

  const bool interior = (x0 >= 0) & (y0 >= 0) & (x0 + K < Wl) &
                        (y0 + K < Hl);
  if (interior) {
    // Fast path (mirror of the forward's): stream patch ROWS —
    // A_j(u) = wx0*g[u][j] + wx1*g[u-1][j]; patch row j = wy0*A_j + wy1*A_{j-1}
    // — the (K+1) stores per row are CONTIGUOUS in the map.
    float aprev[K + 1], acur[K + 1];
#pragma unroll
    for (int j = 0; j <= K; ++j) {
      if (j < K) {
        float grow[K];
#pragma unroll
        for (int a = 0; a < K; ++a)
          grow[a] = (float)gin[(long)(a * K + j) * tap_stride];
#pragma unroll
        for (int u = 0; u <= K; ++u) {
          float v = 0.0f;
          if (u < K) v += wx0 * grow[u];
          if (u > 0) v += wx1 * grow[u - 1];
          acur[u] = v;
        }
      } else {
#pragma unroll
        for (int u = 0; u <= K; ++u) acur[u] = 0.0f;
      }
      scalar_t* dst = gmap + (long)(y0 + j) * Wl + x0;
#pragma unroll
      for (int u = 0; u <= K; ++u) {
        float v = wy0 * acur[u];
        if (j > 0) v += wy1 * aprev[u];
        dst[u] = (scalar_t)(acc ? (float)dst[u] + v : v);
      }
#pragma unroll
      for (int u = 0; u <= K; ++u) aprev[u] = acur[u];
    }
    return;
  }

  // patch[u][j] = sum_{a,c} wx_{u-a} wy_{j-c} g[a][c]; stream over patch
  // columns u holding g columns a=u-1 (gprev) and a=u (gcur) in registers.
  float gprev[K], gcur[K];
#pragma unroll
  for (int c = 0; c < K; ++c) gprev[c] = 0.0f;
#pragma unroll
  for (int a = 0; a <= K; ++a) {
#pragma unroll
    for (int c = 0; c < K; ++c)
      gcur[c] = (a < K) ? (float)gin[(long)(a * K + c) * tap_stride] : 0.0f;

    const int xx = x0 + a;
    if ((xx >= 0) & (xx < Wl)) {
      float tx[K];
#pragma unroll
      for (int c = 0; c < K; ++c) tx[c] = wx1 * gprev[c] + wx0 * gcur[c];
#pragma unroll
      for (int j = 0; j <= K; ++j) {
        const int yy = y0 + j;
        if ((yy >= 0) & (yy < Hl)) {
          float v = 0.0f;
          if (j < K) v += wy0 * tx[j];
          if (j > 0) v += wy1 * tx[j - 1];
          // each patch position of this thread's PRIVATE map is touched
          // exactly once per call: plain store on a fresh buffer, one RMW
          // on the iteration-chained buffer
          scalar_t* d = gmap + (long)yy * Wl + xx;
          *d = (scalar_t)(acc ? (float)*d + v : v);
        }
      }
    }
#pragma unroll
    for (int c = 0; c < K; ++c) gprev[c] = gcur[c];
  }
}

template <int R, typename scalar_t>
static void lookup_fwd_all(const LkLevels& lv, const float* coords,
                           scalar_t* out, int BP, int P, int L, int cl,
                           int ldc, hipStream_t stream) {
  dim3 grid(fh_cdiv(BP, LK_THREADS), L);
  hipLaunchKernelGGL((corr_lookup_fwd_kernel<R, scalar_t>), grid,
                     dim3(LK_THREADS), 0, stream, lv, coords, out, BP, P,
                     L, cl, ldc);
}

template <int R, typename scalar_t>
static void lookup_bwd_all(const scalar_t* gout, const float* coords,
                           const LkLevels& lv, int BP, int P, int L, int cl,
                           int acc, hipStream_t stream) {
  dim3 grid(fh_cdiv(BP, LK_THREADS), L);
  hipLaunchKernelGGL((corr_lookup_bwd_kernel<R, scalar_t>), grid,
                     dim3(LK_THREADS), 0, stream, gout, coords, lv, BP, P,
                     L, cl, acc);
}

// levels may be fp32 (reference parity) or bf16 (HBM-resident bf16
// pyramid); out / gout stay fp32, the bilerp blend always runs fp32.
// out / gout dtype == level dtype (bf16-resident pyramid emits bf16 taps
// straight into the bf16 NHWC motion-encoder conv; fp32 keeps parity).
void flowhip_corr_lookup_fwd_launch(const void* const* levels,
                                    const int* Hs, const int* Ws,
                                    const float* coords,
                                    void* out, int BP, int P,
                                    int L, int radius, int cl, int ldc,
                                    int is_bf16, hipStream_t stream) {
  LkLevels lv{};
  for (int l = 0; l < L; ++l) { lv.p[l] = levels[l]; lv.H[l] = Hs[l]; lv.W[l] = Ws[l]; }
  if (is_bf16) {
    switch (radius) {
      case 3: lookup_fwd_all<3>(lv, coords, (__bf16*)out, BP, P, L, cl, ldc, stream); break;
      case 4: lookup_fwd_all<4>(lv, coords, (__bf16*)out, BP, P, L, cl, ldc, stream); break;
      default: abort();
    }
  } else {
    switch (radius) {
      case 3: lookup_fwd_all<3>(lv, coords, (float*)out, BP, P, L, cl, ldc, stream); break;
      case 4: lookup_fwd_all<4>(lv, coords, (float*)out, BP, P, L, cl, ldc, stream); break;
      default: abort();
    }
  }
}

void flowhip_corr_lookup_bwd_launch(const void* gout, const float* coords,
                                    void* const* glevels, const int* Hs,
                                    const int* Ws, int BP, int P,
                                    int L, int radius, int cl,
                                    int is_bf16, int acc,
                                    hipStream_t stream) {
  LkLevels lv{};
  for (int l = 0; l < L; ++l) { lv.g[l] = glevels[l]; lv.H[l] = Hs[l]; lv.W[l] = Ws[l]; }
  if (is_bf16) {
    switch (radius) {
      case 3: lookup_bwd_all<3>((const __bf16*)gout, coords, lv, BP, P, L, cl, acc, stream); break;
      case 4: lookup_bwd_all<4>((const __bf16*)gout, coords, lv, BP, P, L, cl, acc, stream); break;
      default: abort();
    }
  } else {
    switch (radius) {
      case 3: lookup_bwd_all<3>((const float*)gout, coords, lv, BP, P, L, cl, acc, stream); break;
      case 4: lookup_bwd_all<4>((const float*)gout, coords, lv, BP, P, L, cl, acc, stream); break;
      default: abort();
    }
  }
}
