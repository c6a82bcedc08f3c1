// Halo-staged 3x3 stride-1 implicit-GEMM convolution (the dominant conv
// shape: every encoder stage conv + the update block's 3x3 convs).
//
// The generic conv_gemm kernel restages its A (im2col) tile from global
// memory once per TAP — 9x the input traffic for a 3x3 kernel. Here a
// workgroup stages ONE [TH+2][TW+2][64] channels-last halo tile of the
// input per 64-channel slab and derives all 9 taps' MFMA A-fragments from
// it by shifted LDS reads; the 9 taps' MFMAs then run back-to-back from
// one staged image (one vmcnt/barrier pair per SLAB instead of per tap).
// B (the packed weight [kyx][o][cpad]) is read as direct global b128
// fragments — 8 KB per tap, L1/L2-resident and shared by every workgroup.
//
//   out[n, y, x, o] = act( sum_{ky,kx,c} x[n, y+ky-1, x+kx-1, c] *
//                          wpk[kyx][o][c] + bias[o] )
//
// Tile: TH=4 rows x TW=32 px = 128 output pixels x 64 output channels per
// workgroup (256 threads, 4 waves as 2Mx2N of 64px x 32co). Per-slab LDS:
// 6*34*64 bf16 = 26 KB, double-buffered = 52 KB -> 3 workgroups/CU.
// The XOR bank swizzle lives on the glds SOURCE channel slot (guide §5
// rule 21), keyed by the tile-local pixel index.
//
// Same-kernel backward-data: stride-1 3x3 bwd-data is this conv with the
// flipped/transposed pack (the caller swaps wpk). Cin ragged (%8) reads
// the zero page past Cin like the generic kernel; Cout ragged clamps the
// B row and masks the store.

#include "common.h"

#define CH_TW 32
#define CH_THREADS 256
// LDS tile row stride for kernel width KW: CH_TW + KW - 1
// 16-B pieces per slab tile: (TH+KH-1) * (TW+KW-1) * 8

// conflict-free for the b128 fragment read: 16 consecutive pixels' slot
// indices (pix%2)*8 + slot^((pix>>1)&7) cover all 16 positions of the
// 256-B bank row
__device__ __forceinline__ unsigned ch_swz(unsigned pix, unsigned slot) {
  return slot ^ ((pix >> 1) & 7u);
}

// stage one 64-channel slab of the halo tile; OOB pixels/channels read the
// zero page. Optional second source (virtually-concatenated input).
template <int THT, int KHH, int KWW>
__device__ __forceinline__ void ch_stage(
    const __bf16* __restrict__ x, const __bf16* __restrict__ x2,
    const __bf16* __restrict__ zpage, char* lds_buf, int n, int y0, int x0,
    int H, int W, int ld_x, int ld_x2, int C1, int Cin, int c0) {
  constexpr int LW = CH_TW + KWW - 1;
  constexpr int PIECES = (THT + KHH - 1) * LW * 8;
  for (int piece0 = threadIdx.x; piece0 < PIECES; piece0 += CH_THREADS) {
    const int pix = piece0 >> 3;        // 0 .. LH*LW-1
    const int slot = piece0 & 7;
    const int sslot = ch_swz(pix, slot);
    const int py = pix / LW, px = pix - py * LW;
    const int gy = y0 + py - KHH / 2;
    const int gx = x0 + px - KWW / 2;
    const int c = c0 + sslot * 8;
    const __bf16* src = zpage;
    if (gy >= 0 && gy < H && gx >= 0 && gx < W && c < Cin) {
      if (x2 == nullptr || c < C1)
        src = x + (((long)n * H + gy) * W + gx) * ld_x + c;
      else
        src = x2 + (((long)n * H + gy) * W + gx) * ld_x2 + (c - C1);
    }
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(lds_buf + piece0 * 16),
        16, 0, 0);
  }
}

// THT: output rows per tile. THT=4 double-buffers the slab loop
// (multi-slab Cin); THT=8 single-buffers (Cin <= 64: one slab, nothing to
// overlap — spend the LDS on a taller tile instead, halving the staging
// and barrier cost per output pixel). (KHH, KWW) generalizes the tap
// geometry: (3,3) encoder/update convs, (1,5)/(5,1) the separable GRU
// convs (same one-stage-all-taps property).
template <int ACT, int THT, bool DBUF, int KHH, int KWW>
__global__ __launch_bounds__(CH_THREADS, 4) void conv_halo3_fwd_kernel(
    const __bf16* __restrict__ x, const __bf16* __restrict__ x2,
    const __bf16* __restrict__ wpk,   // (KH*KW, Cout, cpad)
    const float* __restrict__ bias,
    __bf16* __restrict__ out,         // (N*H*W, ldo) channels-last rows
    const __bf16* __restrict__ zpage,
    int N, int H, int W, int ld_x, int ld_x2, int C1, int Cin, int Cout,
    int cpad, int ldo, int ntx, int nty, int nco) {
  constexpr int LW = CH_TW + KWW - 1;
  constexpr int PIECES = (THT + KHH - 1) * LW * 8;
  __shared__ __attribute__((aligned(16))) char lds[(DBUF ? 2 : 1) * PIECES *
                                                   16];

  int t = blockIdx.x;
  const int tx = t % ntx; t /= ntx;
  const int ty = t % nty; t /= nty;
  const int co_blk = t % nco; t /= nco;
  const int n = t;
  const int x0 = tx * CH_TW, y0 = ty * THT;
  const int n0 = co_blk * 64;

  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int wr = (wave >> 1) * (THT * 16);  // pixel offset of the wave
  const int wc = (wave & 1) * 32;           // cout offset (0|32)

  f32x4 acc[THT][2];
#pragma unroll
  for (int i = 0; i < THT; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int cslabs = cpad / 64;
  ch_stage<THT, KHH, KWW>(x, x2, zpage, lds, n, y0, x0, H, W, ld_x, ld_x2,
                          C1, Cin, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  const int frow = lane & 15;  // fragment row (pixel within 16)
  const int fk = lane >> 4;    // k subchunk

  int cur = 0;
  for (int cs = 0; cs < cslabs; ++cs) {
    if (DBUF && cs + 1 < cslabs)
      ch_stage<THT, KHH, KWW>(x, x2, zpage, lds + (cur ^ 1) * PIECES * 16,
                              n, y0, x0, H, W, ld_x, ld_x2, C1, Cin,
                              (cs + 1) * 64);

    const char* abuf = lds + cur * PIECES * 16;
#pragma unroll 1
    for (int kyx = 0; kyx < KHH * KWW; ++kyx) {
      const int ky = kyx / KWW, kx = kyx - ky * KWW;
      const __bf16* wsec = wpk + ((long)kyx * Cout + n0) * cpad + cs * 64;
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
        // B fragments: direct global b128 (weights are L2-hot)
        bf16x8 bfr[2];
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          int o = wc + j * 16 + frow;
          if (n0 + o >= Cout) o = Cout - 1 - n0;  // clamp; store masks
          bfr[j] = *(const bf16x8*)(wsec + (long)o * cpad + kk * 32 +
                                    fk * 8);
        }
#pragma unroll
        for (int i = 0; i < THT; ++i) {
          // A fragment: 16 pixels' tap-shifted 16-B channel chunks
          const int p = wr + i * 16 + frow;           // tile pixel
          const int py = p >> 5, px = p & 31;
          const unsigned pix = (unsigned)((py + ky) * LW + (px + kx));
          const unsigned slot = (unsigned)(kk * 4 + fk);
          const bf16x8 afr = *(const bf16x8*)(
              abuf + pix * 128 + ch_swz(pix, slot) * 16);
#pragma unroll
          for (int j = 0; j < 2; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afr, bfr[j], acc[i][j], 0, 0, 0);
        }
      }
    }
    if (cs + 1 < cslabs) {
      if (!DBUF)
        ch_stage<THT, KHH, KWW>(x, x2, zpage, lds, n, y0, x0, H, W, ld_x,
                                ld_x2, C1, Cin, (cs + 1) * 64);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
      if (DBUF) cur ^= 1;
    }
  }

  // epilogue: bias + activation, masked channels-last stores
  const int fcol = lane & 15;
  const int frow0 = (lane >> 4) * 4;
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    const int gco = n0 + wc + j * 16 + fcol;
    if (gco >= Cout) continue;
    const float b = bias ? bias[gco] : 0.f;
#pragma unroll
    for (int i = 0; i < THT; ++i) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int p = wr + i * 16 + frow0 + r;
        const int py = p >> 5, px = p & 31;
        const int gy = y0 + py, gx = x0 + px;
        if (gy >= H || gx >= W) continue;
        float v = acc[i][j][r] + b;
        if (ACT == 1) v = fmaxf(v, 0.f);
        out[(((long)n * H + gy) * W + gx) * ldo + gco] = (__bf16)v;
      }
    }
  }
}

bool flowhip_conv_halo_fwd_launch(const void* x, const void* x2,
                                  const void* wpk, const float* bias,
                                  void* out, const void* zpage, int N,
                                  int H, int W, int ld_x, int ld_x2, int C1,
                                  int Cin, int Cout, int cpad, int ldo,
                                  int KH, int KW, int act,
                                  hipStream_t stream) {
  if (cpad % 64 != 0) return false;
  const int nco = fh_cdiv(Cout, 64);
  // single-slab inputs (Cin <= 64): taller single-buffered tile
  const int THT = (cpad == 64) ? 8 : 4;
  const int ntx = fh_cdiv(W, CH_TW), nty = fh_cdiv(H, THT);
  const long blocks = (long)ntx * nty * nco * N;
  if (blocks < 320) return false;  // fill-first: generic kernel handles it
  dim3 grid((unsigned)blocks), block(CH_THREADS);
#define CH_LAUNCH(A, T, D, KHH, KWW)                                         \
  hipLaunchKernelGGL((conv_halo3_fwd_kernel<A, T, D, KHH, KWW>), grid,       \
                     block, 0, stream, (const __bf16*)x,                     \
                     (const __bf16*)x2, (const __bf16*)wpk, bias,            \
                     (__bf16*)out, (const __bf16*)zpage, N, H, W, ld_x,      \
                     ld_x2, C1, Cin, Cout, cpad, ldo, ntx, nty, nco)
#define CH_GEOM(KHH, KWW)                                                    \
  if (KH == KHH && KW == KWW) {                                              \
    if (THT == 8) {                                                          \
      if (act == 1) CH_LAUNCH(1, 8, false, KHH, KWW);                        \
      else CH_LAUNCH(0, 8, false, KHH, KWW);                                 \
    } else {                                                                 \
      if (act == 1) CH_LAUNCH(1, 4, true, KHH, KWW);                         \
      else CH_LAUNCH(0, 4, true, KHH, KWW);                                  \
    }                                                                        \
    return true;                                                             \
  }
  CH_GEOM(3, 3) CH_GEOM(1, 5) CH_GEOM(5, 1)
#undef CH_GEOM
#undef CH_LAUNCH
  return false;
}
