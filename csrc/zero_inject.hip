// Sparse zero-injection upsample (kernel #8 of SURVEY.md §2.2; reference
// upsampler.py:179-210 get_out_tensor + strided assignment).
//
//   out[:, :, sH//2::sH, sW//2::sW] = inp ; zeros elsewhere
//
// The torch form allocates + zero-fills the high-res tensor, then runs a
// strided copy (and the autograd backward re-slices) — 3-4 kernels per
// call, 24 calls per training step at full res. This writes the output in
// one coalesced pass; backward is one strided gather.

#include "common.h"

#define ZI_THREADS 256

__global__ __launch_bounds__(ZI_THREADS) void zero_inject_fwd_kernel(
    const float* __restrict__ inp,  // (N, C, ih, iw)
    float* __restrict__ out,        // (N, C, oh, ow)
    long total, int ih, int iw, int oh, int ow, int sH, int sW) {
  const int offH = sH / 2, offW = sW / 2;
  for (long idx = (long)blockIdx.x * ZI_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * ZI_THREADS) {
    long t = idx;
    const int x = t % ow; t /= ow;
    const int y = t % oh; t /= oh;
    const long nc = t;
    float v = 0.0f;
    const int ry = y - offH, rx = x - offW;
    if (ry >= 0 && rx >= 0 && ry % sH == 0 && rx % sW == 0) {
      const int iy = ry / sH, ix = rx / sW;
      if (iy < ih && ix < iw) v = inp[(nc * ih + iy) * iw + ix];
    }
    out[idx] = v;
  }
}

__global__ __launch_bounds__(ZI_THREADS) void zero_inject_bwd_kernel(
    const float* __restrict__ gout,  // (N, C, oh, ow)
    float* __restrict__ dinp,        // (N, C, ih, iw)
    long total, int ih, int iw, int oh, int ow, int sH, int sW) {
  const int offH = sH / 2, offW = sW / 2;
  for (long idx = (long)blockIdx.x * ZI_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * ZI_THREADS) {
    long t = idx;
    const int ix = t % iw; t /= iw;
    const int iy = t % ih; t /= ih;
    const long nc = t;
    dinp[idx] = gout[(nc * oh + offH + (long)iy * sH) * ow + offW +
                     (long)ix * sW];
  }
}

void flowhip_zero_inject_fwd_launch(const float* inp, float* out, long total,
                                    int ih, int iw, int oh, int ow, int sH,
                                    int sW, hipStream_t stream) {
  long blocks = (total + ZI_THREADS - 1) / ZI_THREADS;
  if (blocks > 32768) blocks = 32768;
  hipLaunchKernelGGL(zero_inject_fwd_kernel, dim3((int)blocks),
                     dim3(ZI_THREADS), 0, stream, inp, out, total, ih, iw,
                     oh, ow, sH, sW);
}

void flowhip_zero_inject_bwd_launch(const float* gout, float* dinp,
                                    long total, int ih, int iw, int oh,
                                    int ow, int sH, int sW,
                                    hipStream_t stream) {
  long blocks = (total + ZI_THREADS - 1) / ZI_THREADS;
  if (blocks > 32768) blocks = 32768;
  hipLaunchKernelGGL(zero_inject_bwd_kernel, dim3((int)blocks),
                     dim3(ZI_THREADS), 0, stream, gout, dinp, total, ih, iw,
                     oh, ow, sH, sW);
}
