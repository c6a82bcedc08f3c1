// Pixel-adaptive convolution kernels (#9/#10 of SURVEY.md §2.2; reference
// core/pac_modules.py — NVIDIA PAC). These back the guided-upsampling
// baseline heads (PacJointUpsample / DJIF / JointBilateral,
// pac_upsampler.py); the default NCUP path never calls them
// (ref upsampler.py:12), so they are written as straightforward
// global-gather kernels (L1/L2 do the tap reuse) rather than LDS-tiled.
//
// #9 packernel2d (gaussian, stride 1, channel_wise=False, no mask):
//   u_t(q)  = exp(-0.5 * sum_c (f[c, q+dt] - f[c, q])^2),  dt = dilation*(t - r)
//   k_t(q)  = u_t(q)            (normalize=0)
//           = u_t(q) / sum_j u_j(q)   (normalize=1; 0/0 -> 0 like the ref)
//   backward: dd2_t = -0.5 * u_t * du_t (du via the normalization jacobian
//   when normalize=1); df[c,q] = sum_t 2*dd2_t(q-dt)*(f[c,q]-f[c,q-dt])
//                              - sum_t 2*dd2_t(q)   *(f[c,q+dt]-f[c,q])
//
// #10 pacconv2d (stride 1):
//   out[o,q] = sum_{c,t} x[c, q+dt] * k_t(q) * W[o,c,t] (+ bias)
//   shared_filters: W is (1,1,K,K) and out[c,q] = sum_t x[c,q+dt] k_t(q) W_t
//   backward (dx, dk, dW) by direct gather/reduction.
//
// Layouts: NCHW fp32 contiguous; kernel tensor (B, 1, K, K, H, W).

#include "common.h"

#define PAC_THREADS 256
#define PAC_MAXK2 49

template <int K>
__global__ __launch_bounds__(PAC_THREADS) void packernel_gauss_fwd_kernel(
    const float* __restrict__ f,  // (B, C, H, W)
    float* __restrict__ k,        // (B, K*K, H, W) viewed as (B,1,K,K,H,W)
    long npix, int B, int C, int H, int W, int dil, int norm) {
  constexpr int K2 = K * K;
  const int R = K / 2;
  for (long idx = (long)blockIdx.x * PAC_THREADS + threadIdx.x; idx < npix;
       idx += (long)gridDim.x * PAC_THREADS) {
    long t = idx;
    const int x = t % W; t /= W;
    const int y = t % H; t /= H;
    const long b = t;
    const long plane = (long)H * W;
    const float* fb = f + b * C * plane;

    float u[K2];
    bool valid[K2];
#pragma unroll
    for (int i = 0; i < K2; ++i) u[i] = 0.f;
#pragma unroll
    for (int ky = 0; ky < K; ++ky) {
      const int yy = y + dil * (ky - R);
#pragma unroll
      for (int kx = 0; kx < K; ++kx) {
        const int xx = x + dil * (kx - R);
        valid[ky * K + kx] =
            (yy >= 0) & (yy < H) & (xx >= 0) & (xx < W);
      }
    }

    for (int c = 0; c < C; ++c) {
      const float fc = fb[c * plane + (long)y * W + x];
#pragma unroll
      for (int ky = 0; ky < K; ++ky) {
        const int yy = y + dil * (ky - R);
        const bool vy = (yy >= 0) & (yy < H);
#pragma unroll
        for (int kx = 0; kx < K; ++kx) {
          const int xx = x + dil * (kx - R);
          // out-of-range taps compare against 0 (zero padding, as the
          // reference's nd2col zero-pads the unfolded features)
          const float ft = (vy & (xx >= 0) & (xx < W))
                               ? fb[c * plane + (long)yy * W + xx] : 0.f;
          const float d = ft - fc;
          u[ky * K + kx] += d * d;
        }
      }
    }
    float s = 0.f;
#pragma unroll
    for (int i = 0; i < K2; ++i) {
      u[i] = expf(-0.5f * u[i]);
      // normalize path: the reference multiplies by the unfolded ones
      // mask first, zeroing out-of-range taps (packernel2d :384-389)
      if (norm && !valid[i]) u[i] = 0.f;
      s += u[i];
    }
    const float inv = (norm && s > 0.f) ? 1.0f / s : 1.0f;
    float* kb = k + (b * K2) * plane + (long)y * W + x;
#pragma unroll
    for (int i = 0; i < K2; ++i)
      kb[(long)i * plane] = norm ? u[i] * inv : u[i];
  }
}

// backward: needs f, the SAVED kernel output k, and dk. Emits df.
// For norm=1 the saved k is normalized; recover dd2 via the jacobian:
//   du_t = (dk_t - sum_j dk_j k_j) / s  with s = sum u — we don't store s,
//   so we re-derive: with k_t = u_t/s, du_t = (dk_t - <dk,k>)/s and
//   dd2_t = -0.5 u_t du_t = -0.5 k_t (dk_t - <dk,k>)   (the s cancels).
// For norm=0: dd2_t = -0.5 k_t dk_t.
template <int K>
__global__ __launch_bounds__(PAC_THREADS) void packernel_gauss_bwd_kernel(
    const float* __restrict__ f, const float* __restrict__ k,
    const float* __restrict__ dk, float* __restrict__ df,
    long total, int B, int C, int H, int W, int dil, int norm) {
  constexpr int K2 = K * K;
  const int R = K / 2;
  const long plane = (long)H * W;
  for (long idx = (long)blockIdx.x * PAC_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * PAC_THREADS) {
    long t = idx;
    const int x = t % W; t /= W;
    const int y = t % H; t /= H;
    const int c = t % C; t /= C;
    const long b = t;
    const float* fb = f + (b * C + c) * plane;
    const float* kb = k + b * K2 * plane;
    const float* dkb = dk + b * K2 * plane;

    const float fq = fb[(long)y * W + x];
    float acc = 0.f;

#pragma unroll
    for (int ky = 0; ky < K; ++ky) {
#pragma unroll
      for (int kx = 0; kx < K; ++kx) {
        const int ti = ky * K + kx;
        const int dy = dil * (ky - R), dx = dil * (kx - R);
        // term 1: q is the tap of the window centered at p = q - dt
        {
          const int py = y - dy, px = x - dx;
          if (py >= 0 && py < H && px >= 0 && px < W) {
            const long po = (long)py * W + px;
            float dd2;
            if (norm) {
              float dot = 0.f;
#pragma unroll
              for (int j = 0; j < K2; ++j)
                dot += dkb[(long)j * plane + po] * kb[(long)j * plane + po];
              dd2 = -0.5f * kb[(long)ti * plane + po] *
                    (dkb[(long)ti * plane + po] - dot);
            } else {
              dd2 = -0.5f * kb[(long)ti * plane + po] *
                    dkb[(long)ti * plane + po];
            }
            acc += 2.f * dd2 * (fq - fb[po]);
          }
        }
        // term 2: q is the center of its own window
        {
          const long qo = (long)y * W + x;
          const int ty = y + dy, tx2 = x + dx;
          const float ft = (ty >= 0 && ty < H && tx2 >= 0 && tx2 < W)
                               ? fb[(long)ty * W + tx2] : 0.f;
          float dd2;
          if (norm) {
            float dot = 0.f;
#pragma unroll
            for (int j = 0; j < K2; ++j)
              dot += dkb[(long)j * plane + qo] * kb[(long)j * plane + qo];
            dd2 = -0.5f * kb[(long)ti * plane + qo] *
                  (dkb[(long)ti * plane + qo] - dot);
          } else {
            dd2 = -0.5f * kb[(long)ti * plane + qo] *
                  dkb[(long)ti * plane + qo];
          }
          acc -= 2.f * dd2 * (ft - fq);
        }
      }
    }
    df[idx] = acc;
  }
}

// ---------------------------------------------------------------------------
// #10 pacconv2d, stride 1, same padding (pad = dil*(K-1)/2)
// ---------------------------------------------------------------------------

// General padding/output-size form: out is (B, Co, OH, OW) with
// OH = H + 2*pH - (K-1)*dil + ... (caller computes); tap (ky,kx) of output
// pixel q reads x at (qy - pH + dil*ky, qx - pW + dil*kx); the adapting
// kernel kr lives on the OUTPUT grid.
template <int K>
__global__ __launch_bounds__(PAC_THREADS) void pacconv_fwd_kernel(
    const float* __restrict__ x,   // (B, Ci, H, W)
    const float* __restrict__ kr,  // (B, K2, OH, OW)
    const float* __restrict__ w,   // (Co, Ci, K, K) or (1,1,K,K) shared
    const float* __restrict__ bias,
    float* __restrict__ out,       // (B, Co, OH, OW)
    long total, int B, int Ci, int Co, int H, int W, int OH, int OW, int pH,
    int pW, int dil, int shared) {
  constexpr int K2 = K * K;
  const long plane = (long)H * W;
  const long oplane = (long)OH * OW;
  for (long idx = (long)blockIdx.x * PAC_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * PAC_THREADS) {
    long t = idx;
    const int xq = t % OW; t /= OW;
    const int yq = t % OH; t /= OH;
    const int o = t % Co; t /= Co;
    const long b = t;
    const float* xb = x + b * Ci * plane;
    const float* kb = kr + b * K2 * oplane + (long)yq * OW + xq;

    float acc = bias ? bias[o] : 0.f;
    for (int c = 0; c < Ci; ++c) {
      if (shared && c != o) continue;  // shared filters: diagonal map
      const float* woc = shared ? w : w + ((long)o * Ci + c) * K2;
#pragma unroll
      for (int ky = 0; ky < K; ++ky) {
        const int yy = yq - pH + dil * ky;
        const bool vy = (yy >= 0) & (yy < H);
#pragma unroll
        for (int kx = 0; kx < K; ++kx) {
          const int xx = xq - pW + dil * kx;
          if (vy & (xx >= 0) & (xx < W)) {
            acc += xb[c * plane + (long)yy * W + xx] *
                   kb[(long)(ky * K + kx) * oplane] * woc[ky * K + kx];
          }
        }
      }
    }
    out[idx] = acc;
  }
}

// dx[c,p] = sum_{o,t} dy[o, q] * k_t(q) * W[o,c,t]  over q = p + pH - dil*ky
template <int K>
__global__ __launch_bounds__(PAC_THREADS) void pacconv_bwd_dx_kernel(
    const float* __restrict__ dy, const float* __restrict__ kr,
    const float* __restrict__ w, float* __restrict__ dx,
    long total, int B, int Ci, int Co, int H, int W, int OH, int OW, int pH,
    int pW, int dil, int shared) {
  constexpr int K2 = K * K;
  const long plane = (long)H * W;
  const long oplane = (long)OH * OW;
  for (long idx = (long)blockIdx.x * PAC_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * PAC_THREADS) {
    long t = idx;
    const int xq = t % W; t /= W;
    const int yq = t % H; t /= H;
    const int c = t % Ci; t /= Ci;
    const long b = t;
    const float* dyb = dy + b * Co * oplane;
    const float* kb = kr + b * K2 * oplane;

    float acc = 0.f;
#pragma unroll
    for (int ky = 0; ky < K; ++ky) {
      const int py = yq + pH - dil * ky;
      if (py < 0 || py >= OH) continue;
#pragma unroll
      for (int kx = 0; kx < K; ++kx) {
        const int px = xq + pW - dil * kx;
        if (px < 0 || px >= OW) continue;
        const long po = (long)py * OW + px;
        const float kv = kb[(long)(ky * K + kx) * oplane + po];
        if (shared) {
          acc += dyb[c * oplane + po] * kv * w[ky * K + kx];
        } else {
          for (int o = 0; o < Co; ++o)
            acc += dyb[o * oplane + po] * kv *
                   w[((long)o * Ci + c) * K2 + ky * K + kx];
        }
      }
    }
    dx[idx] = acc;
  }
}

// dk_t(q) = sum_c x[c, q - p + dil*t] * (sum_o dy[o,q] W[o,c,t])
template <int K>
__global__ __launch_bounds__(PAC_THREADS) void pacconv_bwd_dk_kernel(
    const float* __restrict__ dy, const float* __restrict__ x,
    const float* __restrict__ w, float* __restrict__ dk,
    long total, int B, int Ci, int Co, int H, int W, int OH, int OW, int pH,
    int pW, int dil, int shared) {
  constexpr int K2 = K * K;
  const long plane = (long)H * W;
  const long oplane = (long)OH * OW;
  for (long idx = (long)blockIdx.x * PAC_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * PAC_THREADS) {
    long t = idx;
    const int xq = t % OW; t /= OW;
    const int yq = t % OH; t /= OH;
    const int ti = t % K2; t /= K2;
    const long b = t;
    const int ky = ti / K, kx = ti - ky * K;
    const int yy = yq - pH + dil * ky;
    const int xx = xq - pW + dil * kx;
    float acc = 0.f;
    if (yy >= 0 && yy < H && xx >= 0 && xx < W) {
      const float* xb = x + b * Ci * plane + (long)yy * W + xx;
      const float* dyb = dy + b * Co * oplane + (long)yq * OW + xq;
      for (int c = 0; c < Ci; ++c) {
        float wd = 0.f;
        if (shared) {
          wd = dyb[c * oplane] * w[ti];
        } else {
          for (int o = 0; o < Co; ++o)
            wd += dyb[o * oplane] * w[((long)o * Ci + c) * K2 + ti];
        }
        acc += xb[c * plane] * wd;
      }
    }
    dk[idx] = acc;
  }
}

// dW[o,c,t] partials over pixel chunks; final reduce in a second kernel.
template <int K>
__global__ __launch_bounds__(PAC_THREADS) void pacconv_bwd_dw_kernel(
    const float* __restrict__ dy, const float* __restrict__ x,
    const float* __restrict__ kr, float* __restrict__ partials,
    int B, int Ci, int Co, int H, int W, int OH, int OW, int pH, int pW,
    int dil, int shared, int nchunk) {
  constexpr int K2 = K * K;
  const long plane = (long)H * W;
  const long oplane = (long)OH * OW;
  const int nw = shared ? K2 : Co * Ci * K2;
  extern __shared__ float red[];  // nw floats

  const int chunk = blockIdx.x;
  const long total = (long)B * oplane;
  const long p0 = (total * chunk) / nchunk;
  const long p1 = (total * (chunk + 1)) / nchunk;

  for (int i = threadIdx.x; i < nw; i += PAC_THREADS) red[i] = 0.f;
  __syncthreads();

  for (long idx = p0 + threadIdx.x; idx < p1; idx += PAC_THREADS) {
    long t = idx;
    const int xq = t % OW; t /= OW;
    const int yq = t % OH; t /= OH;
    const long b = t;
    const float* xb = x + b * Ci * plane;
    const float* dyb = dy + b * Co * oplane + (long)yq * OW + xq;
    const float* kb = kr + b * K2 * oplane + (long)yq * OW + xq;
#pragma unroll
    for (int ky = 0; ky < K; ++ky) {
      const int yy = yq - pH + dil * ky;
      if (yy < 0 || yy >= H) continue;
#pragma unroll
      for (int kx = 0; kx < K; ++kx) {
        const int xx = xq - pW + dil * kx;
        if (xx < 0 || xx >= W) continue;
        const int ti = ky * K + kx;
        const float kv = kb[(long)ti * oplane];
        for (int c = 0; c < Ci; ++c) {
          const float xv = xb[c * plane + (long)yy * W + xx] * kv;
          if (shared) {
            atomicAdd(&red[ti], dyb[c * oplane] * xv);
          } else {
            for (int o = 0; o < Co; ++o)
              atomicAdd(&red[((long)o * Ci + c) * K2 + ti],
                        dyb[o * oplane] * xv);
          }
        }
      }
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < nw; i += PAC_THREADS)
    partials[(long)chunk * nw + i] = red[i];
}

__global__ __launch_bounds__(PAC_THREADS) void pac_reduce_kernel(
    const float* __restrict__ partials, float* __restrict__ out, int nchunk,
    int nw) {
  for (int i = blockIdx.x * PAC_THREADS + threadIdx.x; i < nw;
       i += gridDim.x * PAC_THREADS) {
    float s = 0.f;
    for (int c = 0; c < nchunk; ++c) s += partials[(long)c * nw + i];
    out[i] = s;
  }
}

// ---------------------------------------------------------------------------

#define PAC_DISPATCH_K(fn, KV, ...)                                           \
  switch (KV) {                                                               \
    case 3: hipLaunchKernelGGL((fn<3>), grid, block, 0, stream, __VA_ARGS__); break; \
    case 5: hipLaunchKernelGGL((fn<5>), grid, block, 0, stream, __VA_ARGS__); break; \
    case 7: hipLaunchKernelGGL((fn<7>), grid, block, 0, stream, __VA_ARGS__); break; \
    default: return false;                                                    \
  }

static inline int pac_blocks(long total) {
  long b = (total + PAC_THREADS - 1) / PAC_THREADS;
  return (int)(b > 16384 ? 16384 : b);
}

bool flowhip_packernel_fwd_launch(const float* f, float* k, int B, int C,
                                  int H, int W, int K, int dil, int norm,
                                  hipStream_t stream) {
  const long npix = (long)B * H * W;
  dim3 grid(pac_blocks(npix)), block(PAC_THREADS);
  PAC_DISPATCH_K(packernel_gauss_fwd_kernel, K, f, k, npix, B, C, H, W, dil,
                 norm)
  return true;
}

bool flowhip_packernel_bwd_launch(const float* f, const float* k,
                                  const float* dk, float* df, int B, int C,
                                  int H, int W, int K, int dil, int norm,
                                  hipStream_t stream) {
  const long total = (long)B * C * H * W;
  dim3 grid(pac_blocks(total)), block(PAC_THREADS);
  PAC_DISPATCH_K(packernel_gauss_bwd_kernel, K, f, k, dk, df, total, B, C, H,
                 W, dil, norm)
  return true;
}

bool flowhip_pacconv_fwd_launch(const float* x, const float* kr,
                                const float* w, const float* bias, float* out,
                                int B, int Ci, int Co, int H, int W, int OH,
                                int OW, int pH, int pW, int K, int dil,
                                int shared, hipStream_t stream) {
  const long total = (long)B * Co * OH * OW;
  dim3 grid(pac_blocks(total)), block(PAC_THREADS);
  PAC_DISPATCH_K(pacconv_fwd_kernel, K, x, kr, w, bias, out, total, B, Ci,
                 Co, H, W, OH, OW, pH, pW, dil, shared)
  return true;
}

bool flowhip_pacconv_bwd_launch(const float* dy, const float* x,
                                const float* kr, const float* w, float* dx,
                                float* dk, float* partials, float* dw,
                                int nchunk, int B, int Ci, int Co, int H,
                                int W, int OH, int OW, int pH, int pW, int K,
                                int dil, int shared, hipStream_t stream) {
  const int K2 = K * K;
  const int nw = shared ? K2 : Co * Ci * K2;
  {
    const long total = (long)B * Ci * H * W;
    dim3 grid(pac_blocks(total)), block(PAC_THREADS);
    PAC_DISPATCH_K(pacconv_bwd_dx_kernel, K, dy, kr, w, dx, total, B, Ci, Co,
                   H, W, OH, OW, pH, pW, dil, shared)
  }
  {
    const long total = (long)B * K2 * OH * OW;
    dim3 grid(pac_blocks(total)), block(PAC_THREADS);
    PAC_DISPATCH_K(pacconv_bwd_dk_kernel, K, dy, x, w, dk, total, B, Ci, Co,
                   H, W, OH, OW, pH, pW, dil, shared)
  }
  {
    dim3 grid(nchunk), block(PAC_THREADS);
    const size_t shmem = (size_t)nw * sizeof(float);
    switch (K) {
      case 3: hipLaunchKernelGGL((pacconv_bwd_dw_kernel<3>), grid, block, shmem, stream, dy, x, kr, partials, B, Ci, Co, H, W, OH, OW, pH, pW, dil, shared, nchunk); break;
      case 5: hipLaunchKernelGGL((pacconv_bwd_dw_kernel<5>), grid, block, shmem, stream, dy, x, kr, partials, B, Ci, Co, H, W, OH, OW, pH, pW, dil, shared, nchunk); break;
      case 7: hipLaunchKernelGGL((pacconv_bwd_dw_kernel<7>), grid, block, shmem, stream, dy, x, kr, partials, B, Ci, Co, H, W, OH, OW, pH, pW, dil, shared, nchunk); break;
      default: return false;
    }
    hipLaunchKernelGGL(pac_reduce_kernel, dim3(fh_cdiv(nw, PAC_THREADS)),
                       block, 0, stream, partials, dw, nchunk, nw);
  }
  return true;
}

// ---------------------------------------------------------------------------
// PAC pooling (kernel #10 tail; reference pac_modules.py:288-329 PacPool2dFn):
//   out[b, c, oy, ox] = sum_{ky,kx} kernel[b, kk(c), ky*K+kx, oy, ox] *
//                       x[b, c, oy*s - p + ky*d, ...]
// kr is passed flattened (B, KCH, K2, OH, OW) with KCH in {1 (shared), C}.
// Runtime K/stride/dilation (baseline-head op, not on the hot path).
// Backward: dx by atomicAdd scatter (overlapping windows under stride <
// K*d), dkernel by per-(b,kch,tap,pixel) gather over channels.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(PAC_THREADS) void pacpool_fwd_kernel(
    const float* __restrict__ x, const float* __restrict__ kr,
    float* __restrict__ out, long total, int B, int C, int KCH, int H, int W,
    int OH, int OW, int K, int sH, int sW, int pH, int pW, int dil) {
  for (long idx = (long)blockIdx.x * PAC_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * PAC_THREADS) {
    long t = idx;
    const int ox = t % OW; t /= OW;
    const int oy = t % OH; t /= OH;
    const int c = t % C; t /= C;
    const int b = (int)t;
    const int kc = KCH == 1 ? 0 : c;
    const float* kb = kr + (((long)b * KCH + kc) * K * K) * OH * OW +
                      (long)oy * OW + ox;
    const float* xb = x + ((long)b * C + c) * H * W;
    float acc = 0.f;
    for (int ky = 0; ky < K; ++ky) {
      const int iy = oy * sH - pH + ky * dil;
      if (iy < 0 || iy >= H) continue;
      for (int kx = 0; kx < K; ++kx) {
        const int ix = ox * sW - pW + kx * dil;
        if (ix < 0 || ix >= W) continue;
        acc += kb[(long)(ky * K + kx) * OH * OW] * xb[(long)iy * W + ix];
      }
    }
    out[idx] = acc;
  }
}

__global__ __launch_bounds__(PAC_THREADS) void pacpool_bwd_dx_kernel(
    const float* __restrict__ dy, const float* __restrict__ kr,
    float* __restrict__ dx, long total, int B, int C, int KCH, int H, int W,
    int OH, int OW, int K, int sH, int sW, int pH, int pW, int dil) {
  for (long idx = (long)blockIdx.x * PAC_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * PAC_THREADS) {
    long t = idx;
    const int ox = t % OW; t /= OW;
    const int oy = t % OH; t /= OH;
    const int c = t % C; t /= C;
    const int b = (int)t;
    const int kc = KCH == 1 ? 0 : c;
    const float g = dy[idx];
    const float* kb = kr + (((long)b * KCH + kc) * K * K) * OH * OW +
                      (long)oy * OW + ox;
    float* xg = dx + ((long)b * C + c) * H * W;
    for (int ky = 0; ky < K; ++ky) {
      const int iy = oy * sH - pH + ky * dil;
      if (iy < 0 || iy >= H) continue;
      for (int kx = 0; kx < K; ++kx) {
        const int ix = ox * sW - pW + kx * dil;
        if (ix < 0 || ix >= W) continue;
        atomicAdd(&xg[(long)iy * W + ix],
                  g * kb[(long)(ky * K + kx) * OH * OW]);
      }
    }
  }
}

__global__ __launch_bounds__(PAC_THREADS) void pacpool_bwd_dk_kernel(
    const float* __restrict__ dy, const float* __restrict__ x,
    float* __restrict__ dk, long total, int B, int C, int KCH, int H, int W,
    int OH, int OW, int K, int sH, int sW, int pH, int pW, int dil) {
  // total = B * KCH * K2 * OH * OW
  const int K2 = K * K;
  for (long idx = (long)blockIdx.x * PAC_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * PAC_THREADS) {
    long t = idx;
    const int ox = t % OW; t /= OW;
    const int oy = t % OH; t /= OH;
    const int kk = t % K2; t /= K2;
    const int kc = t % KCH; t /= KCH;
    const int b = (int)t;
    const int ky = kk / K, kx = kk - ky * K;
    const int iy = oy * sH - pH + ky * dil;
    const int ix = ox * sW - pW + kx * dil;
    float acc = 0.f;
    if (iy >= 0 && iy < H && ix >= 0 && ix < W) {
      const int c0 = KCH == 1 ? 0 : kc;
      const int c1 = KCH == 1 ? C : kc + 1;
      for (int c = c0; c < c1; ++c)
        acc += dy[(((long)b * C + c) * OH + oy) * OW + ox] *
               x[(((long)b * C + c) * H + iy) * W + ix];
    }
    dk[idx] = acc;
  }
}

void flowhip_pacpool_fwd_launch(const float* x, const float* kr, float* out,
                                int B, int C, int KCH, int H, int W, int OH,
                                int OW, int K, int sH, int sW, int pH,
                                int pW, int dil, hipStream_t stream) {
  const long total = (long)B * C * OH * OW;
  dim3 grid(pac_blocks(total)), block(PAC_THREADS);
  hipLaunchKernelGGL(pacpool_fwd_kernel, grid, block, 0, stream, x, kr, out,
                     total, B, C, KCH, H, W, OH, OW, K, sH, sW, pH, pW, dil);
}

void flowhip_pacpool_bwd_launch(const float* dy, const float* x,
                                const float* kr, float* dx, float* dk, int B,
                                int C, int KCH, int H, int W, int OH, int OW,
                                int K, int sH, int sW, int pH, int pW,
                                int dil, hipStream_t stream) {
  {
    const long total = (long)B * C * OH * OW;
    dim3 grid(pac_blocks(total)), block(PAC_THREADS);
    hipLaunchKernelGGL(pacpool_bwd_dx_kernel, grid, block, 0, stream, dy, kr,
                       dx, total, B, C, KCH, H, W, OH, OW, K, sH, sW, pH, pW,
                       dil);
  }
  {
    const long total = (long)B * KCH * K * K * OH * OW;
    dim3 grid(pac_blocks(total)), block(PAC_THREADS);
    hipLaunchKernelGGL(pacpool_bwd_dk_kernel, grid, block, 0, stream, dy, x,
                       dk, total, B, C, KCH, H, W, OH, OW, K, sH, sW, pH, pW,
                       dil);
  }
}
