// Column sum for conv bias gradients: (M, C) bf16 channels-last rows ->
// (C) fp32, dbias[c] = sum_m dy[m, c]. torch's strided reduce over the
// pixel dims of a channels-last tensor runs ~22 us for 5.5 MB (~100 calls
// per step); this is a chunked 16-B-load reduction whose per-chunk rows
// fold into the zero-filled output with one atomic per (chunk, channel).

#include "common.h"

#define CS_THREADS 256
#define CS_MCHUNK 256  // small chunks: enough blocks to fill 256 CUs at M~21k

__global__ __launch_bounds__(CS_THREADS) void col_sum_partial_kernel(
    const __bf16* __restrict__ dy, float* __restrict__ out, long M,
    int C, int nchunk) {
  int b = blockIdx.x;
  const int chunk = b % nchunk; b /= nchunk;
  const int cb = b;
  const int CBW = min(64, C - cb * 64);
  const int CG = CBW / 8;            // 8-channel groups
  const int g = threadIdx.x % CG;
  const int s = threadIdx.x / CG;
  const int S = CS_THREADS / CG;
  const int c0 = cb * 64 + g * 8;

  const long m0 = (long)chunk * CS_MCHUNK;
  const long m1 = min(m0 + CS_MCHUNK, M);

  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;
  if (s < S) {
    for (long m = m0 + s; m < m1; m += S) {
      const bf16x8 v = *(const bf16x8*)(dy + m * C + c0);
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += (float)v[j];
    }
  }

  __shared__ float red[64 * 33];
  for (int j = 0; j < 8; ++j)
    if (s < S && s < 32) red[(g * 8 + j) * 33 + s] = acc[j];
  __syncthreads();
  if (s >= 32 && s < S)
    for (int j = 0; j < 8; ++j)
      atomicAdd(&red[(g * 8 + j) * 33 + (s & 31)], acc[j]);
  __syncthreads();
  if (threadIdx.x < (unsigned)CBW) {
    float t = 0.f;
    const int smax = S < 32 ? S : 32;
    for (int ss = 0; ss < smax; ++ss) t += red[threadIdx.x * 33 + ss];
    // one device-scope atomic per (chunk, channel) straight into the
    // output (out is zero-filled by the caller): drops the separate
    // finalize launch + the partials round trip (~10 us/call of
    // latency-bound reduce across ~160 bias grads per step). Accumulation
    // order across chunks is non-deterministic — like torch's own
    // multi-block reductions; magnitudes are fp32 partial sums of
    // comparable scale.
    atomicAdd(&out[cb * 64 + threadIdx.x], t);
  }
}


// Dual column sum for the frozen-BatchNorm backward: sum_m g[m,c] AND
// sum_m g[m,c]*x[m,c] in one pass over both tensors (torch's
// native_batch_norm_backward spends ~105 us/call on the same reduction
// pair at encoder shapes). out layout: (2, C), zero-filled by the caller.
__global__ __launch_bounds__(CS_THREADS) void col_sum2_partial_kernel(
    const __bf16* __restrict__ g, const __bf16* __restrict__ x,
    float* __restrict__ out, long M, int C, int nchunk) {
  int b = blockIdx.x;
  const int chunk = b % nchunk; b /= nchunk;
  const int cb = b;
  const int CBW = min(64, C - cb * 64);
  const int CG = CBW / 8;
  const int gi = threadIdx.x % CG;
  const int s = threadIdx.x / CG;
  const int S = CS_THREADS / CG;
  const int c0 = cb * 64 + gi * 8;

  const long m0 = (long)chunk * CS_MCHUNK;
  const long m1 = min(m0 + CS_MCHUNK, M);

  float ag[8], agx[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) ag[j] = agx[j] = 0.f;
  if (s < S) {
    for (long m = m0 + s; m < m1; m += S) {
      const bf16x8 vg = *(const bf16x8*)(g + m * C + c0);
      const bf16x8 vx = *(const bf16x8*)(x + m * C + c0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const float fg = (float)vg[j];
        ag[j] += fg;
        agx[j] += fg * (float)vx[j];
      }
    }
  }

  __shared__ float red[2][64 * 33];
  for (int j = 0; j < 8; ++j)
    if (s < S && s < 32) {
      red[0][(gi * 8 + j) * 33 + s] = ag[j];
      red[1][(gi * 8 + j) * 33 + s] = agx[j];
    }
  __syncthreads();
  if (s >= 32 && s < S)
    for (int j = 0; j < 8; ++j) {
      atomicAdd(&red[0][(gi * 8 + j) * 33 + (s & 31)], ag[j]);
      atomicAdd(&red[1][(gi * 8 + j) * 33 + (s & 31)], agx[j]);
    }
  __syncthreads();
  if (threadIdx.x < (unsigned)CBW) {
    const int smax = S < 32 ? S : 32;
    float t0 = 0.f, t1 = 0.f;
    for (int ss = 0; ss < smax; ++ss) {
      t0 += red[0][threadIdx.x * 33 + ss];
      t1 += red[1][threadIdx.x * 33 + ss];
    }
    // atomic chunk reduction straight into the zero-filled (2, C) output
    // (see col_sum_partial_kernel)
    atomicAdd(&out[cb * 64 + threadIdx.x], t0);
    atomicAdd(&out[C + cb * 64 + threadIdx.x], t1);
  }
}


// Per-channel plane reduction on NCHW fp32: out[c] = sum_{b,h,w} x (or
// x*y). Replaces the eager mul+sum pair in the nconv backward's
// ds-through-sum(w) term (full-res (B,2,H,W) tensors, ~48 pairs/step).
__global__ __launch_bounds__(256) void plane_dot_sum_kernel(
    const float* __restrict__ x, const float* __restrict__ y,
    float* __restrict__ out, long P, int C, int pchunks) {
  const int bc = blockIdx.x;
  const int c = bc % C;
  const int chunk = blockIdx.y;
  // chunk bounds rounded to float4 granularity (tail in the last chunk)
  long p0 = ((P * (long)chunk) / pchunks) & ~3L;
  long p1 = (chunk + 1 == pchunks) ? P
                                   : ((P * (long)(chunk + 1)) / pchunks) & ~3L;
  const float* xp = x + (long)bc * P;
  const float* yp = y ? y + (long)bc * P : nullptr;
  float acc = 0.f;
  if ((P & 3) == 0) {  // 16-B aligned planes: float4 stream
    const long p1v = p1 & ~3L;
    for (long p = p0 + threadIdx.x * 4; p < p1v; p += 256 * 4) {
      const float4 vx = *(const float4*)(xp + p);
      if (yp) {
        const float4 vy = *(const float4*)(yp + p);
        acc += vx.x * vy.x + vx.y * vy.y + vx.z * vy.z + vx.w * vy.w;
      } else {
        acc += vx.x + vx.y + vx.z + vx.w;
      }
    }
    const long t = p1v + threadIdx.x;
    if (t < p1) acc += yp ? xp[t] * yp[t] : xp[t];
  } else {
    for (long p = p0 + threadIdx.x; p < p1; p += 256)
      acc += yp ? xp[p] * yp[p] : xp[p];
  }

  __shared__ float red[256];
  red[threadIdx.x] = acc;
  __syncthreads();
#pragma unroll
  for (int s = 128; s > 0; s >>= 1) {
    if ((int)threadIdx.x < s) red[threadIdx.x] += red[threadIdx.x + s];
    __syncthreads();
  }
  if (threadIdx.x == 0) atomicAdd(&out[c], red[0]);
}

void flowhip_plane_dot_sum_launch(const float* x, const float* y, float* out,
                                  long P, int B, int C, hipStream_t stream) {
  int pchunks = (int)((512 + (long)B * C - 1) / ((long)B * C));
  if (pchunks < 1) pchunks = 1;
  if (pchunks > 1024) pchunks = 1024;
  if ((long)pchunks > (P + 1023) / 1024) pchunks = (int)((P + 1023) / 1024);
  if (pchunks < 1) pchunks = 1;
  dim3 grid(B * C, pchunks), block(256);
  hipLaunchKernelGGL(plane_dot_sum_kernel, grid, block, 0, stream, x, y, out,
                     P, C, pchunks);
}

// Frozen-BN affine: y[m,c] = bf16(fp32(x[m,c]) * s[c] + t[c]) on
// channels-last rows. fp32 math + bf16 store puts the quantization point
// exactly where the stock autocast path (fp32 BN output, bf16 cast at the
// next conv) puts it. t == nullptr -> t = 0 (the backward dx = g*s pass).
__global__ __launch_bounds__(256) void frozen_bn_apply_kernel(
    const __bf16* __restrict__ x, __bf16* __restrict__ y,
    const float* __restrict__ s, const float* __restrict__ t, long npieces,
    int cpieces) {
  for (long p = (long)blockIdx.x * 256 + threadIdx.x; p < npieces;
       p += (long)gridDim.x * 256) {
    const int c0 = (int)(p % cpieces) * 8;
    const bf16x8 v = *(const bf16x8*)(x + p * 8);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float r = (float)v[j] * s[c0 + j];
      if (t) r += t[c0 + j];
      o[j] = (__bf16)r;
    }
    *(bf16x8*)(y + p * 8) = o;
  }
}

void flowhip_frozen_bn_apply_launch(const void* x, void* y, const float* s,
                                    const float* t, long M, int C,
                                    hipStream_t stream) {
  const long npieces = M * (C / 8);
  long blocks = (npieces + 255) / 256;
  if (blocks > 8192) blocks = 8192;
  hipLaunchKernelGGL(frozen_bn_apply_kernel, dim3((int)blocks), dim3(256), 0,
                     stream, (const __bf16*)x, (__bf16*)y, s, t, npieces,
                     C / 8);
}

bool flowhip_col_sum2_launch(const void* g, const void* x, float* out,
                             long M, int C, int nchunk,
                             hipStream_t stream) {
  if (C % 8 != 0 || C < 8) return false;
  hipLaunchKernelGGL(col_sum2_partial_kernel, dim3(((C + 63) / 64) * nchunk),
                     dim3(CS_THREADS), 0, stream, (const __bf16*)g,
                     (const __bf16*)x, out, M, C, nchunk);
  return true;
}

bool flowhip_col_sum_launch(const void* dy, float* out,
                            long M, int C, int nchunk, hipStream_t stream) {
  if (C % 8 != 0 || C < 8) return false;
  // `out` must be zero-filled (the binding allocates torch::zeros); the
  // chunk kernel reduces into it atomically — no finalize pass.
  hipLaunchKernelGGL(col_sum_partial_kernel, dim3(((C + 63) / 64) * nchunk),
                     dim3(CS_THREADS), 0, stream, (const __bf16*)dy,
                     out, M, C, nchunk);
  return true;
}
