// Batched bf16 GEMM for the all-pairs correlation volume (kernel #1 of
// SURVEY.md §2.2) and its backward products:
//
//   C[bat] (M,N) fp32 = alpha * A[bat] (M,K) @ B[bat] (N,K)^T
//
// Both operands are row-major, K-contiguous bf16 — chosen so both LDS tiles
// are [row][k] and every MFMA fragment is one swizzled ds_read_b128 of 8
// contiguous k-elements (CDNA4 16x16x32 bf16 MFMA, fp32 accumulate).
//
// Structure (cdna_hip_programming.md §5): 128x128 tile, BK=64, 256 threads
// (4 waves, 2x2 of 64x64), double-buffered LDS filled by
// global_load_lds_dwordx4 (lane-linear image; the XOR bank swizzle lives on
// the per-lane SOURCE address and the fragment-read address — rule 21).
// Swizzle: 16B-slot' = slot ^ ((row>>1)&7): rows of one b128 lane-group land
// on distinct banks (bank = (row*32 + slot'*4) % 64 is injective over 16
// consecutive rows).
//
// Shapes in this framework: forward M=N=P (H/8*W/8), K=D (256|128);
// backward M=P, N=D, K=P (the caller pads K to a multiple of 64 with zeros).
// M and N edges are handled by clamping staging rows (duplicate loads) and
// masking the C store.

#include "common.h"

#define BM 128
#define BN 128
#define BK 64
#define THREADS 256

// LDS pieces: one piece = 16 B = 8 bf16. Tile = BM rows x 8 slots.
#define SLOTS_PER_ROW (BK * 2 / 16)  // 8
#define PIECES_PER_TILE (BM * SLOTS_PER_ROW)  // 1024
#define PIECES_PER_THREAD (PIECES_PER_TILE / THREADS)  // 4

__device__ __forceinline__ unsigned swz_slot(unsigned row, unsigned slot) {
  return slot ^ ((row >> 1) & 7u);
}

// Stage one BM x BK tile into LDS with global_load_lds (16 B per lane).
// The LDS destination of glds is WAVE-UNIFORM base + lane*16 (the hardware
// ignores per-lane LDS addresses), so the base is computed from the wave id
// and the per-lane scatter (including the bank swizzle) lives entirely in
// the SOURCE address (guide §5.4 rule 21).
// src: batch base pointer (bf16), row stride ld (elements), tile origin
// (row0, k0). Rows are clamped to [0, nrows) (duplicates are harmless:
// the C-store masks the edge).
__device__ __forceinline__ void stage_tile(const __bf16* __restrict__ src,
                                           int ld, int nrows, int row0, int k0,
                                           char* lds_buf, int wave, int lane) {
#pragma unroll
  for (int j = 0; j < PIECES_PER_THREAD; ++j) {
    const int piece0 = wave * 64 + THREADS * j;  // wave-uniform
    const int piece = piece0 + lane;
    const int row = piece / SLOTS_PER_ROW;
    const int slot = piece % SLOTS_PER_ROW;
    const int src_slot = swz_slot(row, slot);
    int grow = row0 + row;
    grow = grow < nrows ? grow : nrows - 1;
    const __bf16* gptr = src + (long)grow * ld + k0 + src_slot * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)gptr,
        (__attribute__((address_space(3))) void*)(lds_buf + piece0 * 16),
        16, 0, 0);
  }
}

template <typename out_t>
__global__ __launch_bounds__(THREADS, 2) void flowhip_bgemm_nt(
    const __bf16* __restrict__ A, const __bf16* __restrict__ B,
    out_t* __restrict__ C, float alpha, int M, int N, int K, long strideA,
    long strideB, long strideC, int tiles_m, int tiles_n) {
  // LDS: [2 buffers][A tile | B tile], each tile BM*BK bf16 = 16 KB.
  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM * BK * 2];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;

  const int bat = blockIdx.z;
  int tile_id = blockIdx.x;
  const int tm = tile_id % tiles_m;
  const int tn = tile_id / tiles_m;
  const int m0 = tm * BM;
  const int n0 = tn * BN;

  const __bf16* Ab = A + (long)bat * strideA;
  const __bf16* Bb = B + (long)bat * strideB;
  out_t* Cb = C + (long)bat * strideC;

  // wave -> 64x64 quadrant
  const int wr = (wave >> 1) * 64;  // 0 or 64 within tile (M)
  const int wc = (wave & 1) * 64;   // 0 or 64 within tile (N)

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const unsigned TS = BM * BK * 2;  // tile bytes
  const int ntiles = K / BK;  // caller guarantees K % 64 == 0
  int cur = 0;

  stage_tile(Ab, K, M, m0, 0, lds, wave, lane);
  stage_tile(Bb, K, N, n0, 0, lds + TS, wave, lane);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  for (int t = 0; t < ntiles; ++t) {
    if (t + 1 < ntiles) {
      stage_tile(Ab, K, M, m0, (t + 1) * BK, lds + (cur ^ 1) * 2 * TS, wave, lane);
      stage_tile(Bb, K, N, n0, (t + 1) * BK, lds + (cur ^ 1) * 2 * TS + TS, wave, lane);
    }

    // fragment reads + MFMA over the two 32-deep k-chunks of this tile
    const char* abuf = lds + cur * 2 * TS;
    const char* bbuf = lds + cur * 2 * TS + TS;
    const int frow = lane & 15;       // fragment row/col within 16
    const int fk = (lane >> 4);       // k-subchunk 0..3 (8 elems each)

#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 afrag[4], bfrag[4];
#pragma unroll
      for (int m = 0; m < 4; ++m) {
        const unsigned row = wr + m * 16 + frow;
        const unsigned slot = kk * 4 + fk;
        afrag[m] = *(const bf16x8*)(abuf + row * (BK * 2) +
                                    swz_slot(row, slot) * 16);
      }
#pragma unroll
      for (int n = 0; n < 4; ++n) {
        const unsigned row = wc + n * 16 + frow;
        const unsigned slot = kk * 4 + fk;
        bfrag[n] = *(const bf16x8*)(bbuf + row * (BK * 2) +
                                    swz_slot(row, slot) * 16);
      }
#pragma unroll
      for (int m = 0; m < 4; ++m)
#pragma unroll
        for (int n = 0; n < 4; ++n)
          acc[m][n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afrag[m], bfrag[n], acc[m][n], 0, 0, 0);
    }

    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  // Epilogue: C/D fragment layout for 16x16x32: col = lane&15,
  // row = (lane>>4)*4 + reg. Store masked at the M/N edges.
  const int fcol = lane & 15;
  const int frow0 = (lane >> 4) * 4;
#pragma unroll
  for (int m = 0; m < 4; ++m) {
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int gm = m0 + wr + m * 16 + frow0 + r;
        const int gn = n0 + wc + n * 16 + fcol;
        if (gm < M && gn < N)
          Cb[(long)gm * N + gn] = (out_t)(alpha * acc[m][n][r]);
      }
    }
  }
}

void flowhip_bgemm_nt_launch(const void* A, const void* B, void* C,
                             float alpha, int batch, int M, int N, int K,
                             int out_bf16, hipStream_t stream) {
  const int tiles_m = fh_cdiv(M, BM);
  const int tiles_n = fh_cdiv(N, BN);
  dim3 grid(tiles_m * tiles_n, 1, batch);
  dim3 block(THREADS);
  if (out_bf16)
    hipLaunchKernelGGL(flowhip_bgemm_nt<__bf16>, grid, block, 0, stream,
                       (const __bf16*)A, (const __bf16*)B, (__bf16*)C, alpha,
                       M, N, K, (long)M * K, (long)N * K, (long)M * N,
                       tiles_m, tiles_n);
  else
    hipLaunchKernelGGL(flowhip_bgemm_nt<float>, grid, block, 0, stream,
                       (const __bf16*)A, (const __bf16*)B, (float*)C, alpha,
                       M, N, K, (long)M * K, (long)N * K, (long)M * N,
                       tiles_m, tiles_n);
}
