// LDS-tiled normalized-convolution kernels, v2 (kernels #6/#7 of
// SURVEY.md §2.2; math contract = reference nconv_modules.py:164-199).
//
// Replaces the v1 global-gather kernels in nconv.hip for K in {3,5}: the
// v1 kernels re-read every input element K*K times through L1 and (wrw)
// funnel 200k+ atomics onto ~100 words; rocprofv3 showed nconv_wrw alone
// at 31.7% of a training step (profiles/r01_trace_step.md). v2 stages the
// (conf, data*conf) halo tile through LDS once per workgroup and reduces
// the weight gradient hierarchically: registers -> wave shuffle -> LDS ->
// one partial row per workgroup -> tiny second-stage reduce kernel. No
// atomics anywhere, fully deterministic.
//
// Tile geometry: TW=64 (one wave covers a row), TH=16 output rows per
// workgroup of 256 threads (each thread owns 4 pixels, stride-4 rows).
// LDS per plane pair = (TH+K-1)*(TW+K-1)*2 floats; CI<=2 K=5 ~22 KB,
// CI=4 K=3 ~38 KB -> 4+ workgroups/CU.

#include "common.h"

#define NCT_THREADS 256
#define NCT_TW 64
#define NCT_TH 16

// ---------------------------------------------------------------------------
// Shared tile loader: stages conf and data*conf (with K/2 halo, zero-padded)
// for all CI channels into LDS.  LW = TW+K-1 row stride.
// ---------------------------------------------------------------------------
template <int K, int CI, int TH = NCT_TH>
__device__ inline void nct_stage_tile(const float* __restrict__ data,
                                      const float* __restrict__ conf,
                                      float* __restrict__ lds_c,
                                      float* __restrict__ lds_dc,
                                      int n, int x0, int y0, int Ci_stride_n,
                                      int H, int W) {
  constexpr int LW = NCT_TW + K - 1;
  constexpr int LH = TH + K - 1;
  const long plane = (long)H * W;
  for (int idx = threadIdx.x; idx < LH * LW; idx += NCT_THREADS) {
    const int row = idx / LW, col = idx - row * LW;
    const int gy = y0 - K / 2 + row;
    const int gx = x0 - K / 2 + col;
    const bool in = (gy >= 0 && gy < H && gx >= 0 && gx < W);
    const long goff = (long)gy * W + gx;
#pragma unroll
    for (int ci = 0; ci < CI; ++ci) {
      const long base = ((long)n * Ci_stride_n + ci) * plane;
      float c = 0.f, d = 0.f;
      if (in) {
        c = conf[base + goff];
        d = data[base + goff];
      }
      lds_c[ci * LH * LW + idx] = c;
      lds_dc[ci * LH * LW + idx] = d * c;
    }
  }
}

// ---------------------------------------------------------------------------
// Forward:  out = conv(dc, w) / (conv(c, w) + 1e-20) [+bias]
//           cout = conv(c, w) / sum_w   (confidence propagation)
// Grid: (ntx*nty*N); each thread computes NCT_TH/4 pixels for ALL Co.
// ---------------------------------------------------------------------------
template <int K, int CI, int CO>
__global__ __launch_bounds__(NCT_THREADS) void nconv_fwd_tiled_kernel(
    const float* __restrict__ data, const float* __restrict__ conf,
    const float* __restrict__ weight, const float* __restrict__ bias,
    float* __restrict__ out, float* __restrict__ cout,
    int N, int H, int W, int ntx, int nty) {
  constexpr int LW = NCT_TW + K - 1;
  constexpr int LH = NCT_TH + K - 1;
  __shared__ float lds_c[CI * LH * LW];
  __shared__ float lds_dc[CI * LH * LW];
  __shared__ float wsh[CO * CI * K * K];
  __shared__ float winv[CO];

  int t = blockIdx.x;
  const int tx = t % ntx; t /= ntx;
  const int ty = t % nty; t /= nty;
  const int n = t;
  const int x0 = tx * NCT_TW, y0 = ty * NCT_TH;

  constexpr int nw = CO * CI * K * K;
  for (int i = threadIdx.x; i < nw; i += NCT_THREADS) wsh[i] = weight[i];
  __syncthreads();
  if (threadIdx.x < CO) {
    float s = 0.f;
    for (int i = 0; i < CI * K * K; ++i) s += wsh[threadIdx.x * CI * K * K + i];
    winv[threadIdx.x] = 1.0f / s;
  }
  nct_stage_tile<K, CI>(data, conf, lds_c, lds_dc, n, x0, y0, CI, H, W);
  __syncthreads();

  const long plane = (long)H * W;
  const int lx = threadIdx.x & 63;          // 0..63 within tile row
  const int ly0 = threadIdx.x >> 6;         // 0..3
  const int x = x0 + lx;

#pragma unroll 1
  for (int j = 0; j < NCT_TH / 4; ++j) {
    const int lyy = ly0 + 4 * j;
    const int y = y0 + lyy;
    if (x >= W || y >= H) continue;
    float denom[CO], nomin[CO];
#pragma unroll
    for (int co = 0; co < CO; ++co) { denom[co] = 0.f; nomin[co] = 0.f; }
#pragma unroll
    for (int ci = 0; ci < CI; ++ci) {
      const float* lc = lds_c + ci * LH * LW;
      const float* ldc = lds_dc + ci * LH * LW;
      // bounded unroll (K*CO*2 fma + K*2 LDS loads in flight per row):
      // full K*K unrolling costs 256 VGPRs -> 1 wave/SIMD
#pragma unroll 1
      for (int ky = 0; ky < K; ++ky) {
#pragma unroll
        for (int kx = 0; kx < K; ++kx) {
          const float c = lc[(lyy + ky) * LW + lx + kx];
          const float dc = ldc[(lyy + ky) * LW + lx + kx];
#pragma unroll
          for (int co = 0; co < CO; ++co) {
            const float w = wsh[((co * CI + ci) * K + ky) * K + kx];
            denom[co] += w * c;
            nomin[co] += w * dc;
          }
        }
      }
    }
#pragma unroll
    for (int co = 0; co < CO; ++co) {
      float v = nomin[co] / (denom[co] + 1e-20f);
      if (bias != nullptr) v += bias[co];
      const long o = ((long)n * CO + co) * plane + (long)y * W + x;
      out[o] = v;
      cout[o] = denom[co] * winv[co];
    }
  }
}

// ---------------------------------------------------------------------------
// Backward data: transposed-conv gather of (dnomin, ddenom) staged in LDS.
//   g  = convT(dnomin, w[.,ci]);  gd = convT(ddenom, w[.,ci])
//   ddata = conf * g ;  dconf = data * g + gd
// LDS planes here are the Co gradient channels (gn, gd).
// ---------------------------------------------------------------------------
template <int K, int CO>
__global__ __launch_bounds__(NCT_THREADS) void nconv_bwd_data_tiled_kernel(
    const float* __restrict__ dnomin, const float* __restrict__ ddenom,
    const float* __restrict__ data, const float* __restrict__ conf,
    const float* __restrict__ weight,
    float* __restrict__ ddata, float* __restrict__ dconf,
    int N, int Ci, int H, int W, int ntx, int nty) {
  constexpr int LW = NCT_TW + K - 1;
  constexpr int LH = NCT_TH + K - 1;
  __shared__ float lds_gn[CO * LH * LW];
  __shared__ float lds_gd[CO * LH * LW];
  __shared__ float wsh[8 * CO * K * K];  // [co][ci][ky][kx] layout below

  int t = blockIdx.x;
  const int tx = t % ntx; t /= ntx;
  const int ty = t % nty; t /= nty;
  const int n = t;
  const int x0 = tx * NCT_TW, y0 = ty * NCT_TH;

  const int nw = CO * Ci * K * K;
  for (int i = threadIdx.x; i < nw; i += NCT_THREADS) wsh[i] = weight[i];
  __syncthreads();

  // stage gn, gd (note: "data=dnomin, conf=ddenom" pairing abuse of the
  // generic loader would multiply them; stage directly instead)
  {
    const long plane = (long)H * W;
    for (int idx = threadIdx.x; idx < LH * LW; idx += NCT_THREADS) {
      const int row = idx / LW, col = idx - row * LW;
      const int gy = y0 - K / 2 + row;
      const int gx = x0 - K / 2 + col;
      const bool in = (gy >= 0 && gy < H && gx >= 0 && gx < W);
      const long goff = (long)gy * W + gx;
#pragma unroll
      for (int co = 0; co < CO; ++co) {
        const long base = ((long)n * CO + co) * plane;
        lds_gn[co * LH * LW + idx] = in ? dnomin[base + goff] : 0.f;
        lds_gd[co * LH * LW + idx] = in ? ddenom[base + goff] : 0.f;
      }
    }
  }
  __syncthreads();

  const long plane = (long)H * W;
  const int lx = threadIdx.x & 63;
  const int ly0 = threadIdx.x >> 6;
  const int x = x0 + lx;

#pragma unroll
  for (int j = 0; j < NCT_TH / 4; ++j) {
    const int lyy = ly0 + 4 * j;
    const int y = y0 + lyy;
    if (x >= W || y >= H) continue;
    for (int ci = 0; ci < Ci; ++ci) {
      float g = 0.f, gd = 0.f;
#pragma unroll
      for (int co = 0; co < CO; ++co) {
        const float* lgn = lds_gn + co * LH * LW;
        const float* lgd = lds_gd + co * LH * LW;
        // bounded unroll: full K*K*CO unrolling put 50+ LDS loads in
        // flight and cost 256 VGPRs (1 wave/SIMD); K kx-taps of ILP is
        // plenty for ~50-cycle LDS latency at 4+ waves
#pragma unroll 1
        for (int ky = 0; ky < K; ++ky) {
#pragma unroll
          for (int kx = 0; kx < K; ++kx) {
            // transposed conv: out[y] gathers in[y - ky + K/2]; in LDS
            // coords (center at +K/2): row = lyy + K-1-ky, col = lx + K-1-kx
            const float w = wsh[((co * Ci + ci) * K + ky) * K + kx];
            g += w * lgn[(lyy + K - 1 - ky) * LW + lx + K - 1 - kx];
            gd += w * lgd[(lyy + K - 1 - ky) * LW + lx + K - 1 - kx];
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
}

// ---------------------------------------------------------------------------
// Weight gradient, stage 1: per-workgroup partial sums.
//   dw[co,ci,ky,kx] = sum_p dnomin[co,p] * dc[ci,p+d] + ddenom[co,p] * c[ci,p+d]
// Each thread accumulates the full (CI*K*K) slice for one co at a time in
// registers over its 4 pixels (taps from LDS), then the workgroup reduces
// (wave shuffle -> LDS across waves) and writes one row of partials.
// partials layout: (nblocks, Co*CI*K*K).
// ---------------------------------------------------------------------------
#define NCT_TH_WRW 32

template <int K, int CI>
__global__ __launch_bounds__(NCT_THREADS) void nconv_wrw_tiled_kernel(
    const float* __restrict__ dnomin, const float* __restrict__ ddenom,
    const float* __restrict__ data, const float* __restrict__ conf,
    float* __restrict__ partials,
    int N, int Co, int H, int W, int ntx, int nty) {
  constexpr int LW = NCT_TW + K - 1;
  constexpr int LH = NCT_TH_WRW + K - 1;
  constexpr int NW = CI * K * K;  // weights per co
  __shared__ float lds_c[CI * LH * LW];
  __shared__ float lds_dc[CI * LH * LW];
  __shared__ float red[4 * NW];  // cross-wave reduction buffer

  int t = blockIdx.x;
  const int tx = t % ntx; t /= ntx;
  const int ty = t % nty; t /= nty;
  const int n = t;
  const int x0 = tx * NCT_TW, y0 = ty * NCT_TH_WRW;

  nct_stage_tile<K, CI, NCT_TH_WRW>(data, conf, lds_c, lds_dc, n, x0, y0,
                                    CI, H, W);
  __syncthreads();

  const long plane = (long)H * W;
  const int lx = threadIdx.x & 63;
  const int ly0 = threadIdx.x >> 6;
  const int x = x0 + lx;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;

  for (int co = 0; co < Co; ++co) {
    float acc[NW];
#pragma unroll
    for (int i = 0; i < NW; ++i) acc[i] = 0.f;

#pragma unroll
    for (int j = 0; j < NCT_TH_WRW / 4; ++j) {
      const int lyy = ly0 + 4 * j;
      const int y = y0 + lyy;
      if (x >= W || y >= H) continue;
      const long go = ((long)n * Co + co) * plane + (long)y * W + x;
      const float gn = dnomin[go];
      const float gd = ddenom[go];
#pragma unroll
      for (int ci = 0; ci < CI; ++ci) {
        const float* lc = lds_c + ci * LH * LW;
        const float* ldc = lds_dc + ci * LH * LW;
#pragma unroll
        for (int ky = 0; ky < K; ++ky) {
#pragma unroll
          for (int kx = 0; kx < K; ++kx) {
            const float c = lc[(lyy + ky) * LW + lx + kx];
            const float dc = ldc[(lyy + ky) * LW + lx + kx];
            acc[(ci * K + ky) * K + kx] += gn * dc + gd * c;
          }
        }
      }
    }

    // wave-level shuffle reduction of each accumulator
#pragma unroll
    for (int i = 0; i < NW; ++i) {
      float v = acc[i];
#pragma unroll
      for (int s = 32; s > 0; s >>= 1) v += __shfl_down(v, s, 64);
      if (lane == 0) red[wave * NW + i] = v;
    }
    __syncthreads();
    // first NW threads fold the 4 wave rows and emit the partial row
    if (threadIdx.x < NW) {
      const float v = red[threadIdx.x] + red[NW + threadIdx.x] +
                      red[2 * NW + threadIdx.x] + red[3 * NW + threadIdx.x];
      partials[(long)blockIdx.x * (Co * NW) + co * NW + threadIdx.x] = v;
    }
    __syncthreads();
  }
}

// Stage 2: dw[w] = sum_b partials[b, w].  One workgroup per weight value.
__global__ __launch_bounds__(NCT_THREADS) void nconv_wrw_reduce_kernel(
    const float* __restrict__ partials, float* __restrict__ dweight,
    int nblocks, int nw) {
  const int w = blockIdx.x;
  float s = 0.f;
  for (int b = threadIdx.x; b < nblocks; b += NCT_THREADS)
    s += partials[(long)b * nw + w];
  __shared__ float red[4];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
#pragma unroll
  for (int sh = 32; sh > 0; sh >>= 1) s += __shfl_down(s, sh, 64);
  if (lane == 0) red[wave] = s;
  __syncthreads();
  if (threadIdx.x == 0)
    dweight[w] = red[0] + red[1] + red[2] + red[3];
}

// ---------------------------------------------------------------------------
// Launchers.  K in {3,5}; CI (fwd/wrw) and Co (bwd-data) dispatched over
// {1,2,4,8}.  Returns false if the shape is outside the tiled space (caller
// falls back to the v1 kernels).
// ---------------------------------------------------------------------------

#define NCT_CASE_FWD(KK, CIV, COV)                                             \
  if (K == KK && Ci == CIV && Co == COV) {                                     \
    hipLaunchKernelGGL((nconv_fwd_tiled_kernel<KK, CIV, COV>), grid, block, 0, \
                       stream, data, conf, weight, bias, out, cout, N, H, W,   \
                       ntx, nty);                                              \
    return true;                                                               \
  }

bool flowhip_nconv_fwd_tiled_launch(const float* data, const float* conf,
                                    const float* weight, const float* bias,
                                    float* out, float* cout, int N, int Ci,
                                    int Co, int H, int W, int K,
                                    hipStream_t stream) {
  const int ntx = fh_cdiv(W, NCT_TW), nty = fh_cdiv(H, NCT_TH);
  dim3 grid(ntx * nty * N), block(NCT_THREADS);
  NCT_CASE_FWD(5, 1, 1) NCT_CASE_FWD(5, 1, 2) NCT_CASE_FWD(5, 1, 4)
  NCT_CASE_FWD(5, 2, 1) NCT_CASE_FWD(5, 2, 2) NCT_CASE_FWD(5, 2, 4)
  NCT_CASE_FWD(5, 4, 2) NCT_CASE_FWD(5, 4, 4)
  NCT_CASE_FWD(3, 1, 1) NCT_CASE_FWD(3, 1, 2) NCT_CASE_FWD(3, 1, 4)
  NCT_CASE_FWD(3, 2, 1) NCT_CASE_FWD(3, 2, 2) NCT_CASE_FWD(3, 2, 4)
  NCT_CASE_FWD(3, 4, 2) NCT_CASE_FWD(3, 4, 4) NCT_CASE_FWD(3, 8, 2)
  NCT_CASE_FWD(1, 1, 1) NCT_CASE_FWD(1, 1, 2) NCT_CASE_FWD(1, 2, 1)
  NCT_CASE_FWD(1, 2, 2) NCT_CASE_FWD(1, 4, 2)
  return false;
}

#define NCT_CASE_BWD(KK, COV)                                                  \
  if (K == KK && Co == COV) {                                                  \
    hipLaunchKernelGGL((nconv_bwd_data_tiled_kernel<KK, COV>), grid, block, 0, \
                       stream, dnomin, ddenom, data, conf, weight, ddata,      \
                       dconf, N, Ci, H, W, ntx, nty);                          \
    return true;                                                               \
  }

bool flowhip_nconv_bwd_data_tiled_launch(
    const float* dnomin, const float* ddenom, const float* data,
    const float* conf, const float* weight, float* ddata, float* dconf, int N,
    int Ci, int Co, int H, int W, int K, hipStream_t stream) {
  const int ntx = fh_cdiv(W, NCT_TW), nty = fh_cdiv(H, NCT_TH);
  dim3 grid(ntx * nty * N), block(NCT_THREADS);
  NCT_CASE_BWD(5, 1) NCT_CASE_BWD(5, 2) NCT_CASE_BWD(5, 4)
  NCT_CASE_BWD(3, 1) NCT_CASE_BWD(3, 2) NCT_CASE_BWD(3, 4) NCT_CASE_BWD(3, 8)
  NCT_CASE_BWD(1, 1) NCT_CASE_BWD(1, 2) NCT_CASE_BWD(1, 4)
  return false;
}

#define NCT_CASE_WRW(KK, CIV)                                                  \
  if (K == KK && Ci == CIV) {                                                  \
    hipLaunchKernelGGL((nconv_wrw_tiled_kernel<KK, CIV>), grid, block, 0,      \
                       stream, dnomin, ddenom, data, conf, partials, N, Co, H, \
                       W, ntx, nty);                                           \
    launched = true;                                                           \
  }

int flowhip_nconv_tiled_nblocks(int N, int H, int W) {
  return fh_cdiv(W, NCT_TW) * fh_cdiv(H, NCT_TH_WRW) * N;
}

bool flowhip_nconv_wrw_tiled_launch(const float* dnomin, const float* ddenom,
                                    const float* data, const float* conf,
                                    float* partials, float* dweight, int N,
                                    int Ci, int Co, int H, int W, int K,
                                    hipStream_t stream) {
  const int ntx = fh_cdiv(W, NCT_TW), nty = fh_cdiv(H, NCT_TH_WRW);
  const int nblocks = ntx * nty * N;
  const int nw = Co * Ci * K * K;
  dim3 grid(nblocks), block(NCT_THREADS);
  bool launched = false;
  NCT_CASE_WRW(5, 1) NCT_CASE_WRW(5, 2) NCT_CASE_WRW(5, 4)
  NCT_CASE_WRW(3, 1) NCT_CASE_WRW(3, 2) NCT_CASE_WRW(3, 4) NCT_CASE_WRW(3, 8)
  NCT_CASE_WRW(1, 1) NCT_CASE_WRW(1, 2) NCT_CASE_WRW(1, 4) NCT_CASE_WRW(1, 8)
  if (!launched) return false;
  hipLaunchKernelGGL(nconv_wrw_reduce_kernel, dim3(nw), dim3(NCT_THREADS), 0,
                     stream, partials, dweight, nblocks, nw);
  return true;
}

// ---------------------------------------------------------------------------
// Backward elementwise preamble, fused (was ~7 torch elementwise kernels per
// nconv backward — gout/de, -gout*ratio/de, +gcout/s, contiguous copies):
//   de     = cout * s[co] + eps          (denom reconstructed from cout)
//   dnomin = gout / de
//   ddenom = -gout * (out - bias[co]) / de  [+ gcout / s[co]]
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(NCT_THREADS) void nconv_bwd_prep_kernel(
    const float* __restrict__ gout, const float* __restrict__ gcout,
    const float* __restrict__ out, const float* __restrict__ cout,
    const float* __restrict__ wsum,   // (Co) sum of weights per out channel
    const float* __restrict__ bias,   // (Co) or nullptr
    float* __restrict__ dnomin, float* __restrict__ ddenom,
    long total, long plane, int Co, float eps) {
  for (long idx = (long)blockIdx.x * NCT_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * NCT_THREADS) {
    const int co = (int)((idx / plane) % Co);
    const float s = wsum[co];
    const float go = gout[idx];
    const float de = cout[idx] * s + eps;
    const float inv_de = 1.0f / de;
    const float ratio = bias ? out[idx] - bias[co] : out[idx];
    float dd = -go * ratio * inv_de;
    if (gcout != nullptr) dd += gcout[idx] / s;
    dnomin[idx] = go * inv_de;
    ddenom[idx] = dd;
  }
}

void flowhip_nconv_bwd_prep_launch(const float* gout, const float* gcout,
                                   const float* out, const float* cout,
                                   const float* wsum, const float* bias,
                                   float* dnomin, float* ddenom, long total,
                                   long plane, int Co, float eps,
                                   hipStream_t stream) {
  long blocks = (total + NCT_THREADS - 1) / NCT_THREADS;
  if (blocks > 32768) blocks = 32768;
  hipLaunchKernelGGL(nconv_bwd_prep_kernel, dim3((int)blocks),
                     dim3(NCT_THREADS), 0, stream, gout, gcout, out, cout,
                     wsum, bias, dnomin, ddenom, total, plane, Co, eps);
}
