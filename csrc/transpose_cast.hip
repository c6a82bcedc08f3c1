// Tiled batched transpose + fp32->bf16 cast for the correlation-volume
// backward (functional.py CorrVolumeFn.backward): the eager
// grad.transpose(1,2).to(bf16).contiguous() on the (B,P,P) corr gradient is
// an uncoalesced ~300 us elementwise kernel; this is a 64x64 LDS-staged
// transpose with coalesced loads AND stores (~60 us at 3x7168^2).

#include "common.h"

#define TC_DIM 16   // 16x16 threads
#define TC_TILE 64  // 64x64 tile, each thread 4x4

template <typename in_t>
__global__ __launch_bounds__(TC_DIM * TC_DIM) void transpose_cast_kernel(
    const in_t* __restrict__ in,   // (B, M, N) fp32 or bf16
    __bf16* __restrict__ out,      // (B, N, M)
    int M, int N, int tiles_m) {
  __shared__ __bf16 tile[TC_TILE][TC_TILE + 2];  // +2: bank-conflict pad

  const int tm = (blockIdx.x % tiles_m) * TC_TILE;
  const int tn = (blockIdx.x / tiles_m) * TC_TILE;
  const long base = (long)blockIdx.z * M * N;

  const int tx = threadIdx.x % TC_DIM;  // contiguous dim
  const int ty = threadIdx.x / TC_DIM;

  // load: rows tm+ty+16i, cols tn+tx+16j (coalesced over tx)
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int r = tm + ty + 16 * i;
    if (r >= M) continue;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int c = tn + tx + 16 * j;
      if (c < N)
        tile[ty + 16 * i][tx + 16 * j] =
            (__bf16)in[base + (long)r * N + c];
    }
  }
  __syncthreads();

  // store transposed: out rows = n, cols = m (coalesced over tx)
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int r = tn + ty + 16 * i;  // n index
    if (r >= N) continue;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int c = tm + tx + 16 * j;  // m index
      if (c < M)
        out[base + (long)r * M + c] = tile[tx + 16 * j][ty + 16 * i];
    }
  }
}

void flowhip_transpose_cast_launch(const void* in, void* out, int B, int M,
                                   int N, int in_bf16, hipStream_t stream) {
  const int tiles_m = fh_cdiv(M, TC_TILE);
  const int tiles_n = fh_cdiv(N, TC_TILE);
  dim3 grid(tiles_m * tiles_n, 1, B), block(TC_DIM * TC_DIM);
  if (in_bf16)
    hipLaunchKernelGGL(transpose_cast_kernel<__bf16>, grid, block, 0, stream,
                       (const __bf16*)in, (__bf16*)out, M, N, tiles_m);
  else
    hipLaunchKernelGGL(transpose_cast_kernel<float>, grid, block, 0, stream,
                       (const float*)in, (__bf16*)out, M, N, tiles_m);
}
