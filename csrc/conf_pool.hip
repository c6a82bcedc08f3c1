// Confidence-based 2x pooling (kernel #7 of SURVEY.md §2.2; reference
// nconv_modules.py:94-104 + retrieve_elements_from_indices :19-22):
//
//   conf_ds = max_pool2d(conf, 2, 2) / 4
//   data_ds = data at the argmax-confidence position of each 2x2 block
//
// One forward kernel (writes conf_ds, data_ds, and a 2-bit argmax code per
// output cell) instead of the eager max_pool + div + flatten/gather chain;
// one backward kernel that writes every INPUT position exactly once (reads
// the code; no scatter/atomics). max_pool argmax ties resolve to the first
// (row-major) element, matching torch.

#include "common.h"

#define CP_THREADS 256

__global__ __launch_bounds__(CP_THREADS) void conf_pool_fwd_kernel(
    const float* __restrict__ data, const float* __restrict__ conf,
    float* __restrict__ data_ds, float* __restrict__ conf_ds,
    unsigned char* __restrict__ code, long total, int H, int W, int OH,
    int OW) {
  for (long idx = (long)blockIdx.x * CP_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * CP_THREADS) {
    long t = idx;
    const int ox = t % OW; t /= OW;
    const int oy = t % OH; t /= OH;
    const long nc = t;
    const long base = nc * H * W;
    const int y = 2 * oy, x = 2 * ox;
    // row-major first-max tie-breaking (torch max_pool2d semantics)
    float best = conf[base + (long)y * W + x];
    int arg = 0;
    const bool xv = (x + 1) < W, yv = (y + 1) < H;
    if (xv) {
      const float v = conf[base + (long)y * W + x + 1];
      if (v > best) { best = v; arg = 1; }
    }
    if (yv) {
      const float v = conf[base + (long)(y + 1) * W + x];
      if (v > best) { best = v; arg = 2; }
    }
    if (xv && yv) {
      const float v = conf[base + (long)(y + 1) * W + x + 1];
      if (v > best) { best = v; arg = 3; }
    }
    const long off = base + (long)(y + (arg >> 1)) * W + x + (arg & 1);
    conf_ds[idx] = best * 0.25f;
    data_ds[idx] = data[off];
    code[idx] = (unsigned char)arg;
  }
}

__global__ __launch_bounds__(CP_THREADS) void conf_pool_bwd_kernel(
    const float* __restrict__ gdata_ds, const float* __restrict__ gconf_ds,
    const unsigned char* __restrict__ code, float* __restrict__ gdata,
    float* __restrict__ gconf, long total_in, int H, int W, int OH, int OW) {
  for (long idx = (long)blockIdx.x * CP_THREADS + threadIdx.x;
       idx < total_in; idx += (long)gridDim.x * CP_THREADS) {
    long t = idx;
    const int x = t % W; t /= W;
    const int y = t % H; t /= H;
    const long nc = t;
    float gd = 0.f, gc = 0.f;
    const int oy = y >> 1, ox = x >> 1;
    if (oy < OH && ox < OW) {
      const long o = (nc * OH + oy) * OW + ox;
      const int arg = ((y & 1) << 1) | (x & 1);
      if (code[o] == (unsigned char)arg) {
        gd = gdata_ds ? gdata_ds[o] : 0.f;
        gc = gconf_ds ? gconf_ds[o] * 0.25f : 0.f;
      }
    }
    gdata[idx] = gd;
    gconf[idx] = gc;
  }
}

void flowhip_conf_pool_fwd_launch(const float* data, const float* conf,
                                  float* data_ds, float* conf_ds,
                                  unsigned char* code, long total, int H,
                                  int W, int OH, int OW, hipStream_t stream) {
  long blocks = (total + CP_THREADS - 1) / CP_THREADS;
  if (blocks > 16384) blocks = 16384;
  hipLaunchKernelGGL(conf_pool_fwd_kernel, dim3((int)blocks),
                     dim3(CP_THREADS), 0, stream, data, conf, data_ds,
                     conf_ds, code, total, H, W, OH, OW);
}

void flowhip_conf_pool_bwd_launch(const float* gdata_ds,
                                  const float* gconf_ds,
                                  const unsigned char* code, float* gdata,
                                  float* gconf, long total_in, int H, int W,
                                  int OH, int OW, hipStream_t stream) {
  long blocks = (total_in + CP_THREADS - 1) / CP_THREADS;
  if (blocks > 16384) blocks = 16384;
  hipLaunchKernelGGL(conf_pool_bwd_kernel, dim3((int)blocks),
                     dim3(CP_THREADS), 0, stream, gdata_ds, gconf_ds, code,
                     gdata, gconf, total_in, H, W, OH, OW);
}
