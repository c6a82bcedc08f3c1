// Fused sequence loss + metrics (kernel #12 of SURVEY.md §2.2; reference
// train.py:46-71).
//
//   valid_px = (valid >= 0.5) & (||gt||_2 < max_flow)
//   loss     = sum_i gamma^(n-1-i) * mean(valid_px * |pred_i - gt|)
//   metrics  = EPE mean and 1/3/5px inlier rates of the FINAL prediction
//              over valid_px
//
// The eager-mode chain is ~5 elementwise/reduce kernels per prediction per
// direction (sub, abs, mask-mul, mean; sign/mul/scale in backward) over
// full-res (B,2,H,W) tensors; this reads gt/valid once and every
// prediction once, accumulating per-prediction partial sums in LDS.
// Backward is ONE kernel writing all n prediction grads:
//   dpred_i = gloss * gamma^(n-1-i) / numel * valid_px * sign(pred_i - gt)
//
// Determinism: two-stage reduction (per-block partials -> one finalize
// block), no atomics.

#include "common.h"

#define SL_THREADS 256
#define SL_MAX_PREDS 32
#define SL_BLOCKS 1024

struct SLPtrs {
  const float* p[SL_MAX_PREDS];
};
struct SLGradPtrs {
  float* p[SL_MAX_PREDS];
};

// partials layout: (SL_BLOCKS, n + 5): n masked |.| sums, then
// epe_sum, valid_cnt, c1, c3, c5 of the final prediction.
__global__ __launch_bounds__(SL_THREADS) void seq_loss_fwd_kernel(
    SLPtrs preds, const float* __restrict__ gt,
    const float* __restrict__ valid, float* __restrict__ partials,
    int n, long npix, long plane, float max_flow) {
  extern __shared__ float acc[];  // (n+5) * SL_THREADS
  const int nacc = n + 5;
  for (int j = 0; j < nacc; ++j) acc[j * SL_THREADS + threadIdx.x] = 0.f;

  for (long idx = (long)blockIdx.x * SL_THREADS + threadIdx.x; idx < npix;
       idx += (long)SL_BLOCKS * SL_THREADS) {
    const long b = idx / plane;
    const long p = idx - b * plane;
    const long ou = (b * 2) * plane + p;      // u channel offset
    const long ov = (b * 2 + 1) * plane + p;  // v channel offset
    const float gu = gt[ou], gv = gt[ov];
    const float mag = sqrtf(gu * gu + gv * gv);
    const bool v = (valid[idx] >= 0.5f) & (mag < max_flow);
    if (!v) continue;

    for (int i = 0; i < n; ++i) {
      acc[i * SL_THREADS + threadIdx.x] +=
          fabsf(preds.p[i][ou] - gu) + fabsf(preds.p[i][ov] - gv);
    }
    // final-prediction EPE stats
    const float du = preds.p[n - 1][ou] - gu;
    const float dv = preds.p[n - 1][ov] - gv;
    const float epe = sqrtf(du * du + dv * dv);
    acc[n * SL_THREADS + threadIdx.x] += epe;
    acc[(n + 1) * SL_THREADS + threadIdx.x] += 1.f;
    acc[(n + 2) * SL_THREADS + threadIdx.x] += (epe < 1.f) ? 1.f : 0.f;
    acc[(n + 3) * SL_THREADS + threadIdx.x] += (epe < 3.f) ? 1.f : 0.f;
    acc[(n + 4) * SL_THREADS + threadIdx.x] += (epe < 5.f) ? 1.f : 0.f;
  }
  __syncthreads();

  // block reduction: each accumulator row 256 -> 1
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  __shared__ float wred[5 * 4];
  for (int j0 = 0; j0 < nacc; j0 += 5) {
    const int jend = min(nacc - j0, 5);
    for (int j = 0; j < jend; ++j) {
      float s = acc[(j0 + j) * SL_THREADS + threadIdx.x];
#pragma unroll
      for (int sh = 32; sh > 0; sh >>= 1) s += __shfl_down(s, sh, 64);
      if (lane == 0) wred[j * 4 + wave] = s;
    }
    __syncthreads();
    if (threadIdx.x < jend)
      partials[(long)blockIdx.x * nacc + j0 + threadIdx.x] =
          wred[threadIdx.x * 4 + 0] + wred[threadIdx.x * 4 + 1] +
          wred[threadIdx.x * 4 + 2] + wred[threadIdx.x * 4 + 3];
    __syncthreads();
  }
}

// finalize: out = [loss, epe_mean, f1px, f3px, f5px]
__global__ __launch_bounds__(SL_THREADS) void seq_loss_finalize_kernel(
    const float* __restrict__ partials, float* __restrict__ out, int n,
    float gamma, long numel_full) {
  const int nacc = n + 5;
  __shared__ float sums[SL_MAX_PREDS + 5];
  for (int j = threadIdx.x; j < nacc; j += SL_THREADS) {
    float s = 0.f;
    for (int b = 0; b < SL_BLOCKS; ++b) s += partials[(long)b * nacc + j];
    sums[j] = s;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    float loss = 0.f;
    for (int i = 0; i < n; ++i)
      loss += powf(gamma, (float)(n - 1 - i)) * sums[i] / (float)numel_full;
    const float cnt = sums[n + 1] > 0.f ? sums[n + 1] : 1.f;
    out[0] = loss;
    out[1] = sums[n] / cnt;
    out[2] = sums[n + 2] / cnt;
    out[3] = sums[n + 3] / cnt;
    out[4] = sums[n + 4] / cnt;
  }
}

__global__ __launch_bounds__(SL_THREADS) void seq_loss_bwd_kernel(
    SLPtrs preds, SLGradPtrs grads, const float* __restrict__ gt,
    const float* __restrict__ valid, const float* __restrict__ gloss,
    int n, long npix, long plane, float max_flow, float gamma,
    long numel_full) {
  const float g0 = gloss[0] / (float)numel_full;
  for (long idx = (long)blockIdx.x * SL_THREADS + threadIdx.x; idx < npix;
       idx += (long)gridDim.x * SL_THREADS) {
    const long b = idx / plane;
    const long p = idx - b * plane;
    const long ou = (b * 2) * plane + p;
    const long ov = (b * 2 + 1) * plane + p;
    const float gu = gt[ou], gv = gt[ov];
    const float mag = sqrtf(gu * gu + gv * gv);
    const bool v = (valid[idx] >= 0.5f) & (mag < max_flow);
    for (int i = 0; i < n; ++i) {
      float du = 0.f, dv = 0.f;
      if (v) {
        const float w = powf(gamma, (float)(n - 1 - i)) * g0;
        const float su = preds.p[i][ou] - gu;
        const float sv = preds.p[i][ov] - gv;
        du = w * ((su > 0.f) ? 1.f : ((su < 0.f) ? -1.f : 0.f));
        dv = w * ((sv > 0.f) ? 1.f : ((sv < 0.f) ? -1.f : 0.f));
      }
      grads.p[i][ou] = du;
      grads.p[i][ov] = dv;
    }
  }
}

void flowhip_seq_loss_fwd_launch(const float* const* preds, int n,
                                 const float* gt, const float* valid,
                                 float* partials, float* out, long npix,
                                 long plane, float max_flow, float gamma,
                                 long numel_full, hipStream_t stream) {
  SLPtrs sp;
  for (int i = 0; i < n; ++i) sp.p[i] = preds[i];
  const int nacc = n + 5;
  const size_t shmem = (size_t)nacc * SL_THREADS * sizeof(float);
  hipLaunchKernelGGL(seq_loss_fwd_kernel, dim3(SL_BLOCKS), dim3(SL_THREADS),
                     shmem, stream, sp, gt, valid, partials, n, npix, plane,
                     max_flow);
  hipLaunchKernelGGL(seq_loss_finalize_kernel, dim3(1), dim3(SL_THREADS), 0,
                     stream, partials, out, n, gamma, numel_full);
}

void flowhip_seq_loss_bwd_launch(const float* const* preds, float* const* dst,
                                 int n, const float* gt, const float* valid,
                                 const float* gloss, long npix, long plane,
                                 float max_flow, float gamma, long numel_full,
                                 hipStream_t stream) {
  SLPtrs sp;
  SLGradPtrs gp;
  for (int i = 0; i < n; ++i) {
    sp.p[i] = preds[i];
    gp.p[i] = dst[i];
  }
  long blocks = (npix + SL_THREADS - 1) / SL_THREADS;
  if (blocks > 8192) blocks = 8192;
  hipLaunchKernelGGL(seq_loss_bwd_kernel, dim3((int)blocks), dim3(SL_THREADS),
                     0, stream, sp, gp, gt, valid, gloss, n, npix, plane,
                     max_flow, gamma, numel_full);
}
