// LDS-tiled normalized-convolution kernels, v3 (kernels #6/#7 of
// SURVEY.md §2.2; math contract = reference nconv_modules.py:164-199).
//
// v2 (round 1) staged the (conf, data*conf) halo tile through LDS once per
// workgroup with an atomic-free hierarchical weight-grad reduction, but
// moved every global byte as a SCALAR 4-B-per-lane access and read K*K LDS
// taps per output pixel — profiles put the forward ~2.5x above its HBM
// bound (headroom #5). v3 keeps the structure and:
//   - stages through ALIGNED float4 chunks (16 B/lane; covers the halo by
//     loading [x0-4, x0+TW+4) and scatter-clipping into the LDS tile);
//   - each thread owns 4 ADJACENT output pixels: the K-tap windows of the
//     4 pixels overlap, so one (K+3)-element LDS row read feeds all four
//     (2.5x fewer LDS cycles), and every output store is one float4.
// Requires W % 4 == 0 (all BASELINE/NCUP shapes; others fall back to the
// v1 global-gather kernels in nconv.hip).
//
// Tile geometry: TW=64, TH=16 output rows per 256-thread workgroup
// (fwd/bwd; wrw uses TH=32 in two passes). 16 lanes x 4 px cover a row.

#include "common.h"

#define NCT_THREADS 256
#define NCT_TW 64
#define NCT_TH 16
#define NCT_TH_WRW 32

// ---------------------------------------------------------------------------
// Vectorized halo stage: loads aligned float4 chunks of planes (a, b) and
// stores (a, MUL ? b*a : b) into the LDS tile pair. LW = TW+K-1 row
// stride; chunks cover [x0-4, x0+TW+4) so every halo column lands.
// ---------------------------------------------------------------------------
template <int K, int CH, int TH, bool MUL>
__device__ inline void nct_stage_v(const float* __restrict__ a,
                                   const float* __restrict__ b,
                                   float* __restrict__ lds_a,
                                   float* __restrict__ lds_b,
                                   int n, int x0, int y0, int H, int W) {
  constexpr int LW = NCT_TW + K - 1;
  constexpr int LH = TH + K - 1;
  constexpr int CHUNKS = (NCT_TW + 8) / 4;  // 18 aligned float4 per row
  const long plane = (long)H * W;
  for (int u = threadIdx.x; u < LH * CHUNKS; u += NCT_THREADS) {
    const int row = u / CHUNKS, k = u - row * CHUNKS;
    const int gy = y0 - K / 2 + row;
    const int gx = x0 - 4 + k * 4;
    const bool rin = (gy >= 0 && gy < H);
    const int lc0 = gx - (x0 - K / 2);
#pragma unroll
    for (int ch = 0; ch < CH; ++ch) {
      const long base = ((long)n * CH + ch) * plane + (long)gy * W;
      float4 va = {0.f, 0.f, 0.f, 0.f}, vb = {0.f, 0.f, 0.f, 0.f};
      if (rin) {
        if (gx >= 0 && gx + 3 < W) {
          va = *(const float4*)(a + base + gx);
          vb = *(const float4*)(b + base + gx);
        } else {
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            const int xx = gx + e;
            if (xx >= 0 && xx < W) {
              (&va.x)[e] = a[base + xx];
              (&vb.x)[e] = b[base + xx];
            }
          }
        }
      }
      float* la = lds_a + ch * LH * LW + row * LW;
      float* lb = lds_b + ch * LH * LW + row * LW;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const int lcol = lc0 + e;
        if (lcol >= 0 && lcol < LW) {
          const float av = (&va.x)[e];
          la[lcol] = av;
          lb[lcol] = MUL ? (&vb.x)[e] * av : (&vb.x)[e];
        }
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Forward:  out = conv(dc, w) / (conv(c, w) + 1e-20) [+bias]
//           cout = conv(c, w) / sum_w   (confidence propagation)
// Grid: (ntx*nty*N); thread = 4 adjacent pixels of one row, all Co.
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
  nct_stage_v<K, CI, NCT_TH, true>(conf, data, lds_c, lds_dc, n, x0, y0, H,
                                   W);
  __syncthreads();

  const long plane = (long)H * W;
  const int lxb = (threadIdx.x & 15) * 4;   // tile-local x of pixel 0
  const int row = threadIdx.x >> 4;         // 0..15
  const int x = x0 + lxb;
  const int y = y0 + row;
  if (x >= W || y >= H) return;  // W%4==0: the 4-chunk is all-in or all-out

  float denom[CO][4], nomin[CO][4];
#pragma unroll
  for (int co = 0; co < CO; ++co)
#pragma unroll
    for (int e = 0; e < 4; ++e) { denom[co][e] = 0.f; nomin[co][e] = 0.f; }

#pragma unroll
  for (int ci = 0; ci < CI; ++ci) {
    const float* lc = lds_c + ci * LH * LW;
    const float* ldc = lds_dc + ci * LH * LW;
#pragma unroll 1
    for (int ky = 0; ky < K; ++ky) {
      // one (K+3)-wide row read feeds all 4 pixels' kx windows
      float cbuf[K + 3], dcbuf[K + 3];
      const float* rc = lc + (row + ky) * LW + lxb;
      const float* rdc = ldc + (row + ky) * LW + lxb;
#pragma unroll
      for (int tap = 0; tap < K + 3; ++tap) {
        cbuf[tap] = rc[tap];
        dcbuf[tap] = rdc[tap];
      }
#pragma unroll
      for (int kx = 0; kx < K; ++kx) {
#pragma unroll
        for (int co = 0; co < CO; ++co) {
          const float w = wsh[((co * CI + ci) * K + ky) * K + kx];
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            denom[co][e] += w * cbuf[kx + e];
            nomin[co][e] += w * dcbuf[kx + e];
          }
        }
      }
    }
  }
#pragma unroll
  for (int co = 0; co < CO; ++co) {
    const float b = bias != nullptr ? bias[co] : 0.f;
    float4 vo, vc;
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      (&vo.x)[e] = nomin[co][e] / (denom[co][e] + 1e-20f) + b;
      (&vc.x)[e] = denom[co][e] * winv[co];
    }
    const long o = ((long)n * CO + co) * plane + (long)y * W + x;
    *(float4*)(out + o) = vo;
    *(float4*)(cout + o) = vc;
  }
}

// ---------------------------------------------------------------------------
// Backward data: transposed-conv gather of (dnomin, ddenom) staged in LDS.
//   g  = convT(dnomin, w[.,ci]);  gd = convT(ddenom, w[.,ci])
//   ddata = conf * g ;  dconf = data * g + gd
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
  __shared__ float wsh[8 * CO * K * K];  // [co][ci][ky][kx]

  int t = blockIdx.x;
  const int tx = t % ntx; t /= ntx;
  const int ty = t % nty; t /= nty;
  const int n = t;
  const int x0 = tx * NCT_TW, y0 = ty * NCT_TH;

  const int nw = CO * Ci * K * K;
  for (int i = threadIdx.x; i < nw; i += NCT_THREADS) wsh[i] = weight[i];
  __syncthreads();
  nct_stage_v<K, CO, NCT_TH, false>(dnomin, ddenom, lds_gn, lds_gd, n, x0,
                                    y0, H, W);
  __syncthreads();

  const long plane = (long)H * W;
  const int lxb = (threadIdx.x & 15) * 4;
  const int row = threadIdx.x >> 4;
  const int x = x0 + lxb;
  const int y = y0 + row;
  if (x >= W || y >= H) return;

  for (int ci = 0; ci < Ci; ++ci) {
    float g[4] = {0.f, 0.f, 0.f, 0.f}, gd[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int co = 0; co < CO; ++co) {
      const float* lgn = lds_gn + co * LH * LW;
      const float* lgd = lds_gd + co * LH * LW;
#pragma unroll 1
      for (int ky = 0; ky < K; ++ky) {
        // transposed conv: out[x+e] gathers in[x+e + K-1-kx - (K-1)/2...]
        // in LDS coords: row = row + K-1-ky, cols = lxb + e + K-1-kx
        float nbuf[K + 3], dbuf[K + 3];
        const float* rn = lgn + (row + K - 1 - ky) * LW + lxb;
        const float* rd = lgd + (row + K - 1 - ky) * LW + lxb;
#pragma unroll
        for (int tap = 0; tap < K + 3; ++tap) {
          nbuf[tap] = rn[tap];
          dbuf[tap] = rd[tap];
        }
#pragma unroll
        for (int kx = 0; kx < K; ++kx) {
          const float w = wsh[((co * Ci + ci) * K + ky) * K + kx];
#pragma unroll
          for (int e = 0; e < 4; ++e) {
            g[e] += w * nbuf[K - 1 - kx + e];
            gd[e] += w * dbuf[K - 1 - kx + e];
          }
        }
      }
    }
    const long p = ((long)n * Ci + ci) * plane + (long)y * W + x;
    const float4 c4 = *(const float4*)(conf + p);
    const float4 d4 = *(const float4*)(data + p);
    float4 vd, vc;
#pragma unroll
    for (int e = 0; e < 4; ++e) {
      (&vd.x)[e] = (&c4.x)[e] * g[e];
      (&vc.x)[e] = (&d4.x)[e] * g[e] + gd[e];
    }
    *(float4*)(ddata + p) = vd;
    *(float4*)(dconf + p) = vc;
  }
}

// ---------------------------------------------------------------------------
// Weight gradient, stage 1: per-workgroup partial sums.
//   dw[co,ci,ky,kx] = sum_p dnomin[co,p] * dc[ci,p+d] + ddenom[co,p] * c[ci,p+d]
// Thread accumulates the (CI*K*K) slice for one co at a time over its 4
// adjacent pixels x 2 row passes; hierarchical reduction (wave shuffle ->
// LDS across waves) emits one partial row per workgroup.
// partials layout: (nblocks, Co*CI*K*K).
// ---------------------------------------------------------------------------
// wrw tile height by channel count: as tall as the LDS budget allows —
// the per-co wave-shuffle reduction (6 shfl per accumulator) amortizes
// over TH*TW pixels, so taller tiles cut its share 2-4x.
template <int CI, int K>
constexpr int nct_th_wrw() {
  return CI == 1 ? 64 : CI == 2 ? 48 : CI == 4 ? 32 : 16;
}

template <int K, int CI>
__global__ __launch_bounds__(NCT_THREADS) void nconv_wrw_tiled_kernel(
    const float* __restrict__ dnomin, const float* __restrict__ ddenom,
    const float* __restrict__ data, const float* __restrict__ conf,
    float* __restrict__ partials,
    int N, int Co, int H, int W, int ntx, int nty) {
  constexpr int TH = nct_th_wrw<CI, K>();
  constexpr int LW = NCT_TW + K - 1;
  constexpr int LH = TH + K - 1;
  constexpr int NW = CI * K * K;  // weights per co
  __shared__ float lds_c[CI * LH * LW];
  __shared__ float lds_dc[CI * LH * LW];
  __shared__ float red[4 * NW];

  int t = blockIdx.x;
  const int tx = t % ntx; t /= ntx;
  const int ty = t % nty; t /= nty;
  const int n = t;
  const int x0 = tx * NCT_TW, y0 = ty * TH;

  nct_stage_v<K, CI, TH, true>(conf, data, lds_c, lds_dc, n, x0, y0,
                               H, W);
  __syncthreads();

  const long plane = (long)H * W;
  const int lxb = (threadIdx.x & 15) * 4;
  const int row0 = threadIdx.x >> 4;  // 0..15; two row passes cover TH=32
  const int x = x0 + lxb;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;

  for (int co = 0; co < Co; ++co) {
    float acc[NW];
#pragma unroll
    for (int i = 0; i < NW; ++i) acc[i] = 0.f;

#pragma unroll
    for (int j = 0; j < TH / 16; ++j) {
      const int lyy = row0 + 16 * j;
      const int y = y0 + lyy;
      if (x >= W || y >= H) continue;
      const long go = ((long)n * Co + co) * plane + (long)y * W + x;
      const float4 gn4 = *(const float4*)(dnomin + go);
      const float4 gd4 = *(const float4*)(ddenom + go);
#pragma unroll
      for (int ci = 0; ci < CI; ++ci) {
        const float* lc = lds_c + ci * LH * LW;
        const float* ldc = lds_dc + ci * LH * LW;
#pragma unroll 1
        for (int ky = 0; ky < K; ++ky) {
          float cbuf[K + 3], dcbuf[K + 3];
          const float* rc = lc + (lyy + ky) * LW + lxb;
          const float* rdc = ldc + (lyy + ky) * LW + lxb;
#pragma unroll
          for (int tap = 0; tap < K + 3; ++tap) {
            cbuf[tap] = rc[tap];
            dcbuf[tap] = rdc[tap];
          }
#pragma unroll
          for (int kx = 0; kx < K; ++kx) {
            float v = 0.f;
#pragma unroll
            for (int e = 0; e < 4; ++e)
              v += (&gn4.x)[e] * dcbuf[kx + e] + (&gd4.x)[e] * cbuf[kx + e];
            acc[(ci * K + ky) * K + kx] += v;
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
// Launchers.  K in {1,3,5}; CI (fwd/wrw) and Co (bwd-data) dispatched over
// {1,2,4,8}.  Returns false if the shape is outside the tiled space — W
// not a multiple of 4 (float4 rows) included — and the caller falls back
// to the v1 kernels.
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
  if (W % 4 != 0) return false;
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
  if (W % 4 != 0) return false;
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

static int nct_th_wrw_rt(int Ci) {
  return Ci == 1 ? 64 : Ci == 2 ? 48 : Ci == 4 ? 32 : 16;
}

int flowhip_nconv_tiled_nblocks(int N, int H, int W, int Ci) {
  return fh_cdiv(W, NCT_TW) * fh_cdiv(H, nct_th_wrw_rt(Ci)) * N;
}

bool flowhip_nconv_wrw_tiled_launch(const float* dnomin, const float* ddenom,
                                    const float* data, const float* conf,
                                    float* partials, float* dweight, int N,
                                    int Ci, int Co, int H, int W, int K,
                                    hipStream_t stream) {
  if (W % 4 != 0) return false;
  const int ntx = fh_cdiv(W, NCT_TW), nty = fh_cdiv(H, nct_th_wrw_rt(Ci));
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
