// Fused convex-combination x8 upsample (kernel #11 of SURVEY.md §2.2) —
// the RAFT baseline's learned upsampler (reference core/raft.py:73-84).
//
//   out[n, c, 8y+sy, 8x+sx] = sum_k softmax(mask[n, :, sy, sx, y, x])[k]
//                                   * 8*flow[n, c, y+ky-1, x+kx-1]
// with k = ky*3+kx over the 3x3 neighborhood (zero-padded, F.unfold order)
// and mask channel layout ch = (k*8 + sy)*8 + sx (the reference's
// view(N,1,9,8,8,H,W)).
//
// The reference materializes softmax(mask) (N,9,64,H,W) and the unfold
// product at full resolution; this kernel reads mask+flow once and writes
// the output once. Backward: dmask is per-output-pixel exclusive (no
// atomics); dflow gathers 576 contributions per coarse pixel via fp32
// atomicAdd (each add L2-local).

#include "common.h"

#define CU_THREADS 256

template <int F>
__global__ __launch_bounds__(CU_THREADS) void convex_up_fwd_kernel(
    const float* __restrict__ flow,  // (N, 2, H, W)
    const float* __restrict__ mask,  // (N, 9*F*F, H, W)
    float* __restrict__ out,         // (N, 2, F*H, F*W)
    int N, int H, int W) {
  const long total = (long)N * F * H * F * W;
  const long idx0 = (long)blockIdx.x * CU_THREADS + threadIdx.x;
  if (idx0 >= total) return;

  // decode (n, oy, ox); adjacent threads = adjacent ox (coalesced store)
  long t = idx0;
  const int ox = t % (F * W); t /= (F * W);
  const int oy = t % (F * H); t /= (F * H);
  const int n = t;

  const int x = ox / F, sx = ox % F;
  const int y = oy / F, sy = oy % F;
  const long P = (long)H * W;

  // softmax over the 9 mask logits for this subpixel
  const float* mbase = mask + ((long)n * 9 * F * F) * P + (long)y * W + x;
  float logits[9];
  float mx = -1e30f;
#pragma unroll
  for (int k = 0; k < 9; ++k) {
    logits[k] = mbase[(long)((k * F + sy) * F + sx) * P];
    mx = fmaxf(mx, logits[k]);
  }
  float wsum = 0.f;
#pragma unroll
  for (int k = 0; k < 9; ++k) {
    logits[k] = __expf(logits[k] - mx);
    wsum += logits[k];
  }
  const float inv = 1.0f / wsum;

  const float* fbase = flow + (long)n * 2 * P;
  float acc0 = 0.f, acc1 = 0.f;
#pragma unroll
  for (int ky = 0; ky < 3; ++ky) {
#pragma unroll
    for (int kx = 0; kx < 3; ++kx) {
      const int yy = y + ky - 1, xx = x + kx - 1;
      if (yy < 0 || yy >= H || xx < 0 || xx >= W) continue;
      const float w = logits[ky * 3 + kx] * inv;
      acc0 += w * fbase[(long)yy * W + xx];
      acc1 += w * fbase[P + (long)yy * W + xx];
    }
  }
  float* obase = out + (long)n * 2 * F * H * F * W;
  obase[(long)oy * F * W + ox] = 8.0f * acc0;
  obase[(long)F * H * F * W + (long)oy * F * W + ox] = 8.0f * acc1;
}

template <int F>
__global__ __launch_bounds__(CU_THREADS) void convex_up_bwd_kernel(
    const float* __restrict__ gout,  // (N, 2, F*H, F*W)
    const float* __restrict__ flow,  // (N, 2, H, W)
    const float* __restrict__ mask,  // (N, 9*F*F, H, W)
    float* __restrict__ gflow,       // (N, 2, H, W) zero-init, atomic
    float* __restrict__ gmask,       // (N, 9*F*F, H, W)
    int N, int H, int W) {
  const long total = (long)N * F * H * F * W;
  const long idx0 = (long)blockIdx.x * CU_THREADS + threadIdx.x;
  if (idx0 >= total) return;

  long t = idx0;
  const int ox = t % (F * W); t /= (F * W);
  const int oy = t % (F * H); t /= (F * H);
  const int n = t;

  const int x = ox / F, sx = ox % F;
  const int y = oy / F, sy = oy % F;
  const long P = (long)H * W;

  const float* mbase = mask + ((long)n * 9 * F * F) * P + (long)y * W + x;
  float w[9];
  float mx = -1e30f;
#pragma unroll
  for (int k = 0; k < 9; ++k) {
    w[k] = mbase[(long)((k * F + sy) * F + sx) * P];
    mx = fmaxf(mx, w[k]);
  }
  float wsum = 0.f;
#pragma unroll
  for (int k = 0; k < 9; ++k) {
    w[k] = __expf(w[k] - mx);
    wsum += w[k];
  }
  const float inv = 1.0f / wsum;
#pragma unroll
  for (int k = 0; k < 9; ++k) w[k] *= inv;

  const float* fbase = flow + (long)n * 2 * P;
  const float* gbase = gout + (long)n * 2 * F * H * F * W;
  const float g0 = 8.0f * gbase[(long)oy * F * W + ox];
  const float g1 = 8.0f * gbase[(long)F * H * F * W + (long)oy * F * W + ox];

  // dw_k = sum_c g_c * f_k[c];  dm_k = w_k * (dw_k - sum_j w_j dw_j)
  float dw[9];
  float dot = 0.f;
#pragma unroll
  for (int ky = 0; ky < 3; ++ky) {
#pragma unroll
    for (int kx = 0; kx < 3; ++kx) {
      const int k = ky * 3 + kx;
      const int yy = y + ky - 1, xx = x + kx - 1;
      float f0 = 0.f, f1 = 0.f;
      if (yy >= 0 && yy < H && xx >= 0 && xx < W) {
        f0 = fbase[(long)yy * W + xx];
        f1 = fbase[P + (long)yy * W + xx];
        // dflow: scatter the weighted grad into the neighborhood
        atomicAdd((float*)&gflow[(long)n * 2 * P + (long)yy * W + xx],
                  w[k] * g0);
        atomicAdd((float*)&gflow[(long)n * 2 * P + P + (long)yy * W + xx],
                  w[k] * g1);
      }
      dw[k] = g0 * f0 + g1 * f1;
      dot += w[k] * dw[k];
    }
  }
  float* gm = gmask + ((long)n * 9 * F * F) * P + (long)y * W + x;
#pragma unroll
  for (int k = 0; k < 9; ++k)
    gm[(long)((k * F + sy) * F + sx) * P] = w[k] * (dw[k] - dot);
}

void flowhip_convex_up_fwd_launch(const float* flow, const float* mask,
                                  float* out, int N, int H, int W, int factor,
                                  hipStream_t stream) {
  if (factor != 8) abort();
  const long total = (long)N * 8 * H * 8 * W;
  dim3 grid((unsigned)((total + CU_THREADS - 1) / CU_THREADS));
  hipLaunchKernelGGL((convex_up_fwd_kernel<8>), grid, dim3(CU_THREADS), 0,
                     stream, flow, mask, out, N, H, W);
}

void flowhip_convex_up_bwd_launch(const float* gout, const float* flow,
                                  const float* mask, float* gflow,
                                  float* gmask, int N, int H, int W,
                                  int factor, hipStream_t stream) {
  if (factor != 8) abort();
  const long total = (long)N * 8 * H * 8 * W;
  dim3 grid((unsigned)((total + CU_THREADS - 1) / CU_THREADS));
  hipLaunchKernelGGL((convex_up_bwd_kernel<8>), grid, dim3(CU_THREADS), 0,
                     stream, gout, flow, mask, gflow, gmask, N, H, W);
}
