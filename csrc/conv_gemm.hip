// Implicit-GEMM NHWC bf16 convolution family (kernel #5 of SURVEY.md §2.2)
// for the iterative update block (motion encoder, SepConvGRU, flow head —
// reference update.py:6-146) and the feature/context encoders (reference
// extractor.py:118-193). Dilation 1, odd or 1xK/Kx1 kernels, Cin % 8 == 0
// (callers pad the one 324-channel case).
//
// Tile geometry is templated:
//   BM=64  (4 waves, 256 thr)  — update-block shapes (M ~ 21k, fill first)
//   BM=128 (8 waves, 512 thr)  — encoder shapes (M up to ~1M at 448x1024
//          batch 3): halves the per-output barrier/staging overhead and the
//          B-operand traffic; this is the profile-driven fix for the 19%
//          MFMA-util reading of the round-1 64x64-only kernel
//          (profiles/README.md headroom #2/#4).
// Both stage the im2col gather through LDS with 16-byte global_load_lds
// (lane-linear image, XOR bank swizzle on the SOURCE address — guide §5
// rule 21), double-buffered, bias + ReLU fused into the epilogue.
//
// Stride support (SMODE template):
//   0: stride 1, same padding (the original fast path; src dims == out dims)
//   1: strided direct conv     sy = oy*sH + ky - padH    (encoder stems /
//      downsample convs, stride 2)
//   2: strided TRANSPOSED conv (the backward-data of mode 1): with the
//      flipped/transposed weight pack, t = iy + ky + (padH - KH + 1) taps
//      dy at oy = t/sH iff t >= 0 && t % sH == 0 && oy < srcH.
//      For sH == 1 this degenerates to the classic flipped-conv identity
//      used by the stride-1 backward.
//
//   fwd:       out[m, o] = act( sum_{ky,kx,c} x[m+off, c] * w[o, ky,kx, c] + b[o] )
//   bwd-data:  SAME kernel with w' = flip(w).T packed as [kyx][ci][co]
//   wrw:       dW[o, kyx, c] = sum_m dy[m, o] * x[m*s+off, c]
//              (split-M partials + small reduce; MFMA over the m axis with
//              LDS-transposed fragment reads)
//
// Weight packing (host side, cached per weight version):
//   wpk[kyx][o][c_pad]  (c_pad = roundup(Cin, 64), zero-filled)
// Out-of-bounds taps and the c >= Cin tail read a 16-B zero page through
// the per-lane glds source address.

#include "common.h"

#define CG_BM 64
#define CG_BN 64
#define CG_BK 64
#define CG_THREADS 256

__device__ __forceinline__ unsigned cg_swz(unsigned row, unsigned slot) {
  return slot ^ ((row >> 1) & 7u);
}

// Stage the A (im2col) tile: rows = BM consecutive output pixels, k = one
// 64-channel slab of one (ky,kx) section. OOB rows read the zero page.
// Two-source form: channels [0, C1) come from x, [C1, Cin) from x2 — the
// virtually-concatenated GRU input cat([h, x]) without materializing the
// cat (C1 must be a multiple of 64 so a slab never straddles sources).
// SMODE semantics in the header comment; (dy, dx) carry the per-mode tap
// offset, (srcH, srcW) the source image dims, (HH, WW) the m-mapping dims.
// per-thread pixel coordinates of this block's A rows: computed ONCE per
// block (the m-range is subtile-invariant; recomputing the 64-bit
// divisions per piece per subtile was a measured ~5% of the step)
template <int BM, int THREADS>
__device__ __forceinline__ void cg_coords(
    int wave, int lane, long m0, long Mtot, int HH, int WW,
    long* pn, int* pyy, int* pxx) {
  constexpr int PPT = BM * 8 / THREADS;
#pragma unroll
  for (int j = 0; j < PPT; ++j) {
    const int piece = wave * 64 + THREADS * j + lane;
    long m = m0 + (piece >> 3);
    if (m >= Mtot) m = Mtot - 1;
    pxx[j] = (int)(m % WW);
    pyy[j] = (int)((m / WW) % HH);
    pn[j] = m / ((long)WW * HH);
  }
}

template <int BM, int THREADS, int SMODE>
__device__ __forceinline__ void cg_stage_a(
    const __bf16* __restrict__ x, const __bf16* __restrict__ x2,
    const __bf16* __restrict__ zpage,
    char* lds_buf, int wave, int lane, const long* pn, const int* pyy,
    const int* pxx, int srcH, int srcW, int sH, int sW, int ld_x, int ld_x2,
    int C1, int Cin, int dy, int dx, int c0) {
  constexpr int PPT = BM * 8 / THREADS;
#pragma unroll
  for (int j = 0; j < PPT; ++j) {
    const int piece0 = wave * 64 + THREADS * j;
    const int piece = piece0 + lane;
    const int row = piece >> 3;          // 0..BM-1
    const int slot = piece & 7;
    const int sslot = cg_swz(row, slot);
    const int xx = pxx[j];
    const int yy = pyy[j];
    const long n = pn[j];
    int sy, sx;
    bool ok;
    if (SMODE == 0) {
      sy = yy + dy; sx = xx + dx;
      ok = sy >= 0 && sy < srcH && sx >= 0 && sx < srcW;
    } else if (SMODE == 1) {
      sy = yy * sH + dy; sx = xx * sW + dx;
      ok = sy >= 0 && sy < srcH && sx >= 0 && sx < srcW;
    } else {
      const int ty = yy + dy, tx = xx + dx;
      sy = ty / sH; sx = tx / sW;
      ok = ty >= 0 && tx >= 0 && ty % sH == 0 && tx % sW == 0 &&
           sy < srcH && sx < srcW;
    }
    const int c = c0 + sslot * 8;
    const __bf16* src = zpage;
    if (ok && c < Cin) {
      if (x2 == nullptr || c < C1)
        src = x + (((long)n * srcH + sy) * srcW + sx) * ld_x + c;
      else
        src = x2 + (((long)n * srcH + sy) * srcW + sx) * ld_x2 + (c - C1);
    }
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(lds_buf + piece0 * 16),
        16, 0, 0);
  }
}

// Stage the B (weight) tile: rows = 64 output channels (clamped), k = the
// same 64-channel slab. wsec = wpk section base [kyx][.][.].
template <int THREADS>
__device__ __forceinline__ void cg_stage_b(const __bf16* __restrict__ wsec,
                                           char* lds_buf, int wave, int lane,
                                           int n0, int Cout, int cpad,
                                           int c0) {
  constexpr int PPT = CG_BN * 8 / THREADS;
#pragma unroll
  for (int j = 0; j < PPT; ++j) {
    const int piece0 = wave * 64 + THREADS * j;
    if (piece0 >= CG_BN * 8) break;
    const int piece = piece0 + lane;
    const int row = piece >> 3;
    const int slot = piece & 7;
    const int sslot = cg_swz(row, slot);
    int o = n0 + row;
    if (o >= Cout) o = Cout - 1;
    const __bf16* src = wsec + (long)o * cpad + c0 + sslot * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) void*)src,
        (__attribute__((address_space(3))) void*)(lds_buf + piece0 * 16),
        16, 0, 0);
  }
}

// ACT: 0 = none, 1 = ReLU. BM in {64, 128}: 64 -> 4 waves in a 2Mx2N
// quadrant grid, 128 -> 8 waves as 4Mx2N.
template <int ACT, int BM, int SMODE>
__global__
__launch_bounds__(BM == 64 ? 256 : 512, BM == 64 ? 2 : 4)
void conv_gemm_fwd_kernel(
    const __bf16* __restrict__ x,     // (N, srcH, srcW, ld_x) NHWC rows
    const __bf16* __restrict__ x2,    // second input source or nullptr
    const __bf16* __restrict__ wpk,   // (KYX, Cout, cpad)
    const float* __restrict__ bias,   // (Cout) or nullptr
    __bf16* __restrict__ out,         // (Mtot, Cout) — or (Mtot, osplit)
    __bf16* __restrict__ out2,        // (Mtot, Cout-osplit) or nullptr
    const __bf16* __restrict__ zpage,
    long Mtot, int HH, int WW, int srcH, int srcW, int sH, int sW, int ld_x,
    int ld_x2, int C1, int Cin, int Cout, int cpad, int KH, int KW,
    int offH, int offW, int osplit, int tiles_m) {
  constexpr int THREADS = BM == 64 ? 256 : 512;
  constexpr unsigned TSA = BM * CG_BK * 2;       // A tile bytes
  constexpr unsigned TSB = CG_BN * CG_BK * 2;    // B tile bytes
  __shared__ __attribute__((aligned(16))) char lds[2 * (TSA + TSB)];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;

  const int tm = blockIdx.x % tiles_m;
  const int tn = blockIdx.x / tiles_m;
  const long m0 = (long)tm * BM;
  const int n0 = tn * CG_BN;

  // wave quadrant: BM=64 -> 2Mx2N of 32x32; BM=128 -> 4Mx2N
  const int wr = (BM == 64 ? (wave >> 1) : (wave & 3)) * 32;
  const int wc = (BM == 64 ? (wave & 1) : (wave >> 2)) * 32;

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int cslabs = cpad / CG_BK;
  const int nsub = KH * KW * cslabs;

  long pn[BM * 8 / THREADS];
  int pyy[BM * 8 / THREADS], pxx[BM * 8 / THREADS];
  cg_coords<BM, THREADS>(wave, lane, m0, Mtot, HH, WW, pn, pyy, pxx);

  // subtile s -> (kyx = s / cslabs, c0 = (s % cslabs) * 64)
  cg_stage_a<BM, THREADS, SMODE>(x, x2, zpage, lds, wave, lane, pn, pyy,
                                 pxx, srcH, srcW, sH, sW, ld_x, ld_x2, C1,
                                 Cin, offH, offW, 0);
  cg_stage_b<THREADS>(wpk, lds + TSA, wave, lane, n0, Cout, cpad, 0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (int s = 0; s < nsub; ++s) {
    if (s + 1 < nsub) {
      const int kyx = (s + 1) / cslabs;
      const int cs = (s + 1) - kyx * cslabs;
      const int ky = kyx / KW, kx = kyx - ky * KW;
      cg_stage_a<BM, THREADS, SMODE>(
          x, x2, zpage, lds + (cur ^ 1) * (TSA + TSB), wave, lane, pn, pyy,
          pxx, srcH, srcW, sH, sW, ld_x, ld_x2, C1, Cin, ky + offH,
          kx + offW, cs * CG_BK);
      cg_stage_b<THREADS>(wpk + (long)kyx * Cout * cpad,
                          lds + (cur ^ 1) * (TSA + TSB) + TSA, wave, lane,
                          n0, Cout, cpad, cs * CG_BK);
    }

    const char* abuf = lds + cur * (TSA + TSB);
    const char* bbuf = abuf + TSA;
    const int frow = lane & 15;
    const int fk = lane >> 4;

#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[2], bf[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const unsigned row = wr + i * 16 + frow;
        const unsigned slot = kk * 4 + fk;
        af[i] = *(const bf16x8*)(abuf + row * (CG_BK * 2) +
                                 cg_swz(row, slot) * 16);
      }
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const unsigned row = wc + j * 16 + frow;
        const unsigned slot = kk * 4 + fk;
        bf[j] = *(const bf16x8*)(bbuf + row * (CG_BK * 2) +
                                 cg_swz(row, slot) * 16);
      }
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bf[j],
                                                              acc[i][j], 0,
                                                              0, 0);
    }

    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  // epilogue: bias + activation, bf16 store (m, Cout), edge-masked
  const int fcol = lane & 15;
  const int frow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int gco = n0 + wc + j * 16 + fcol;
      if (gco >= Cout) continue;
      const float b = bias ? bias[gco] : 0.f;
      // split store: channels [0, osplit) -> out, rest -> out2 (the
      // backward-data of a virtually-concatenated input)
      __bf16* dst = out;
      int co = gco, ldo = out2 ? osplit : Cout;
      if (out2 != nullptr && gco >= osplit) {
        dst = out2;
        co = gco - osplit;
        ldo = Cout - osplit;
      }
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long gm = m0 + wr + i * 16 + frow0 + r;
        if (gm >= Mtot) continue;
        float v = acc[i][j][r] + b;
        if (ACT == 1) v = fmaxf(v, 0.f);
        dst[gm * ldo + co] = (__bf16)v;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Weight gradient: dWp[o, kyx, c] partial over an M chunk.
// grid: (tiles_o * tiles_c, KYX, CHUNKS); the GEMM contracts over m, so
// both operands need TRANSPOSED (k=m-major) fragments from tiles whose
// global layout is m-major. gfx950's ds_read_b64_tr_b16 does that
// transpose in hardware: each 16-lane group reads one [4 m][16 col] block
// lane-linearly and receives per lane the 4 m-values of its own column
// (guide T10). The LDS image is therefore laid out as 64 subblocks of
// [4 m][16 c] halfwords (cb-major, then mq), with the tile's m rows
// PERMUTED (wrw_row_to_m) so that the two tr reads of a fragment deliver
// exactly the MFMA a/b-operand k-order (k = (lane>>4)*8 + e). The glds
// staging writes the permuted image directly — the per-piece SOURCE
// address does the permutation, the LDS destination stays lane-linear.
// Replaces the round-1 per-element u16 fragment reads (~13% MFMA util,
// profiles headroom #3).
// partials layout: (CHUNKS, KYX, tiles_o*64, cpad) fp32.
// Strided conv support: m walks dy's (OH, OW) grid, x is gathered at
// sy = yy*sH + tap (x dims srcH, srcW).
// ---------------------------------------------------------------------------

typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;

// image row (0..63) -> tile m row: per 32-row half, rows [q*4..q*4+4) hold
// m = (q&3)*8 + (q>>2)*4 + e so a 16-row tr window yields k-consecutive
// 8-element fragments
__device__ __forceinline__ int wrw_row_to_m(int row) {
  const int half = row & 32;
  const int q = (row & 31) >> 2;
  return half + ((q & 3) << 3) + ((q >> 2) << 2) + (row & 3);
}

__device__ __forceinline__ bf16x4 wrw_tr_read(const char* buf,
                                              unsigned byte_off) {
  bf16x4 v;
  const __attribute__((address_space(3))) char* p =
      (const __attribute__((address_space(3))) char*)(buf + byte_off);
  asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(v) : "v"(p));
  return v;
}

__global__ __launch_bounds__(CG_THREADS, 2) void conv_gemm_wrw_kernel(
    const __bf16* __restrict__ dy,  // (Mtot, Cout)
    const __bf16* __restrict__ x,   // (N, srcH, srcW, ld_x)
    const __bf16* __restrict__ x2,  // second source or nullptr
    float* __restrict__ partials,
    float* __restrict__ biasp,  // (nchunk, tiles_o*64) or nullptr
    const __bf16* __restrict__ zpage,
    long Mtot, int HH, int WW, int srcH, int srcW, int sH, int sW, int ld_x,
    int ld_x2, int C1, int Cin, int Cout, int cpad, int KH, int KW, int padH,
    int padW, int tiles_o, int tiles_c, int nchunk) {
  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * CG_BM * CG_BK * 2];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;

  const int to = blockIdx.x % tiles_o;
  const int tc = blockIdx.x / tiles_o;
  const int kyx = blockIdx.y;
  const int chunk = blockIdx.z;
  const int ky = kyx / KW, kx = kyx - ky * KW;
  const int dyo = ky - padH, dxo = kx - padW;

  const int o0 = to * 64;
  const int c0 = tc * 64;

  const long mtiles = (Mtot + CG_BM - 1) / CG_BM;
  const long t0 = (mtiles * chunk) / nchunk;
  const long t1 = (mtiles * (chunk + 1)) / nchunk;

  const int wr = (wave >> 1) * 32;  // o offset
  const int wc = (wave & 1) * 32;   // c offset

  f32x4 acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // bias gradient for free: dbias[o] = sum_m dy[m, o], and the (tc==0,
  // kyx==0) blocks' even waves already hold every dy fragment exactly
  // once — summing the a-operand elements as they pass gives the
  // per-chunk column sums with no extra launch and no extra global reads
  // (replaces a separate col_sum + torch.zeros pair per conv backward).
  const bool do_bias = biasp != nullptr && tc == 0 && kyx == 0 && wc == 0;
  float bacc[2] = {0.f, 0.f};

  const unsigned TS = CG_BM * CG_BK * 2;
  // piece -> permuted-image coordinates: subblock sb = cb*16 + mq holds
  // [4 m][16 col] halfwords at sb*128 B; piece = sb*8 + pr*2 + ch covers
  // row pr, columns cb*16 + ch*8 .. +8 of image row mq*4 + pr.
  auto stage_dy = [&](long mt, char* buf) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int piece0 = wave * 64 + CG_THREADS * j;
      const int piece = piece0 + lane;
      const int sb = piece >> 3;
      const int pr = (piece >> 1) & 3;
      const int ch = piece & 1;
      const int irow = (sb & 15) * 4 + pr;
      long m = mt * CG_BM + wrw_row_to_m(irow);
      const __bf16* src;
      const int o = o0 + (sb >> 4) * 16 + ch * 8;
      if (m < Mtot && o < Cout)
        src = dy + m * Cout + o;
      else
        src = zpage;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(buf + piece0 * 16),
          16, 0, 0);
    }
  };
  // incremental tile-base coordinates: one 64-bit decomposition of the
  // first staged m, then +CG_BM carries per k-tile — the per-piece
  // m%W / m/W divisions were ~30 runtime instructions each on the
  // staging critical path
  long base_m = t0 * CG_BM;
  int bx = 0, by = 0;
  long bn = 0;
  if (t0 < t1) {
    bx = (int)(base_m % WW);
    by = (int)((base_m / WW) % HH);
    bn = base_m / ((long)WW * HH);
  }
  auto advance_base = [&]() {
    base_m += CG_BM;
    bx += CG_BM;
    while (bx >= WW) {
      bx -= WW;
      if (++by == HH) { by = 0; ++bn; }
    }
  };
  auto stage_x = [&](char* buf) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      const int piece0 = wave * 64 + CG_THREADS * j;
      const int piece = piece0 + lane;
      const int sb = piece >> 3;
      const int pr = (piece >> 1) & 3;
      const int ch = piece & 1;
      const int irow = (sb & 15) * 4 + pr;
      const int dm = wrw_row_to_m(irow);
      const long m = base_m + dm;
      const __bf16* src = zpage;
      if (m < Mtot) {
        int xx = bx + dm, yy = by;
        long n = bn;
        while (xx >= WW) {
          xx -= WW;
          if (++yy == HH) { yy = 0; ++n; }
        }
        const int sy = yy * sH + dyo, sx = xx * sW + dxo;
        const int c = c0 + (sb >> 4) * 16 + ch * 8;
        if (sy >= 0 && sy < srcH && sx >= 0 && sx < srcW && c < Cin) {
          if (x2 == nullptr || c < C1)
            src = x + (((long)n * srcH + sy) * srcW + sx) * ld_x + c;
          else
            src = x2 + (((long)n * srcH + sy) * srcW + sx) * ld_x2 + (c - C1);
        }
      }
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) void*)src,
          (__attribute__((address_space(3))) void*)(buf + piece0 * 16),
          16, 0, 0);
    }
  };

  if (t0 < t1) {
    stage_dy(t0, lds);
    stage_x(lds + TS);
    advance_base();
  }
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();

  int cur = 0;
  for (long t = t0; t < t1; ++t) {
    if (t + 1 < t1) {
      stage_dy(t + 1, lds + (cur ^ 1) * 2 * TS);
      stage_x(lds + (cur ^ 1) * 2 * TS + TS);
      advance_base();
    }

    // hardware-transposed fragment reads (ds_read_b64_tr_b16): fragment
    // (i, kk) = two tr windows of the permuted image — lane receives its
    // own column's m-run (k = (lane>>4)*8 + e) directly in MFMA order.
    // Window (cb, kk, w) sits at byte offset (cb*16 + kk*8 + w*4)*128;
    // lane-linear +lane*8 addresses are conflict-free (banks 2l, 2l+1).
    const char* dbuf = lds + cur * 2 * TS;
    const char* xbuf = dbuf + TS;
    const unsigned lb = (unsigned)lane * 8;

#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x4 a0[2], a1[2], b0[2], b1[2];
#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const unsigned cb = (unsigned)(wr >> 4) + i;  // o column block
        a0[i] = wrw_tr_read(dbuf, (cb * 16 + kk * 8 + 0) * 128 + lb);
        a1[i] = wrw_tr_read(dbuf, (cb * 16 + kk * 8 + 4) * 128 + lb);
      }
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const unsigned cb = (unsigned)(wc >> 4) + j;  // c column block
        b0[j] = wrw_tr_read(xbuf, (cb * 16 + kk * 8 + 0) * 128 + lb);
        b1[j] = wrw_tr_read(xbuf, (cb * 16 + kk * 8 + 4) * 128 + lb);
      }
      // hipcc does not track asm loads: drain lgkm with every destination
      // tied, then fence the (register-only) MFMAs below the wait (§5.4
      // rule 18)
      asm volatile("s_waitcnt lgkmcnt(0)"
                   : "+v"(a0[0]), "+v"(a0[1]), "+v"(a1[0]), "+v"(a1[1]),
                     "+v"(b0[0]), "+v"(b0[1]), "+v"(b1[0]), "+v"(b1[1])
                   :: "memory");
      __builtin_amdgcn_sched_barrier(0);

#pragma unroll
      for (int i = 0; i < 2; ++i) {
        const bf16x8 af = __builtin_shufflevector(a0[i], a1[i], 0, 1, 2, 3,
                                                  4, 5, 6, 7);
        if (do_bias) {
#pragma unroll
          for (int e = 0; e < 8; ++e) bacc[i] += (float)af[e];
        }
#pragma unroll
        for (int j = 0; j < 2; ++j) {
          const bf16x8 bf = __builtin_shufflevector(b0[j], b1[j], 0, 1, 2,
                                                    3, 4, 5, 6, 7);
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf,
                                                              acc[i][j], 0,
                                                              0, 0);
        }
      }
    }

    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    cur ^= 1;
  }

  // store partial tile: partials[chunk][kyx][o0+...][c0+...]
  const int KYX = KH * KW;
  const int orows = tiles_o * 64;
  float* pbase = partials +
                 (((long)chunk * KYX + kyx) * orows) * cpad;
  const int fcol = lane & 15;
  const int frow0 = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
#pragma unroll
    for (int j = 0; j < 2; ++j) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int o = o0 + wr + i * 16 + frow0 + r;
        const int c = c0 + wc + j * 16 + fcol;
        pbase[(long)o * cpad + c] = acc[i][j][r];
      }
    }
  }

  if (do_bias) {
    // a-fragment lane l holds o = l&15; lane groups l>>4 hold disjoint
    // m runs — xor-fold them, lane<16 owns the per-chunk column sum
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      float s = bacc[i];
      s += __shfl_xor(s, 16, 64);
      s += __shfl_xor(s, 32, 64);
      if (lane < 16)
        biasp[(long)chunk * (tiles_o * 64) + o0 + wr + i * 16 + lane] = s;
    }
  }
}

// reduce partials over chunks and scatter into dW (Cout, Cin, KH, KW) fp32
__global__ __launch_bounds__(CG_THREADS) void conv_gemm_wrw_reduce_kernel(
    const float* __restrict__ partials, float* __restrict__ dw,
    const float* __restrict__ biasp, float* __restrict__ dbias,
    int nchunk, int KYX, int orows, int cpad, int Cout, int Cin) {
  if (dbias)
    for (int o = blockIdx.x * CG_THREADS + threadIdx.x; o < Cout;
         o += gridDim.x * CG_THREADS) {
      float s = 0.f;
      for (int ch = 0; ch < nchunk; ++ch) s += biasp[(long)ch * orows + o];
      dbias[o] = s;
    }
  const long total = (long)Cout * Cin * KYX;
  for (long idx = (long)blockIdx.x * CG_THREADS + threadIdx.x; idx < total;
       idx += (long)gridDim.x * CG_THREADS) {
    long t = idx;
    const int kyx = (int)(t % KYX); t /= KYX;
    const int c = (int)(t % Cin); t /= Cin;
    const int o = (int)t;
    float s = 0.f;
    for (int ch = 0; ch < nchunk; ++ch)
      s += partials[((((long)ch * KYX + kyx) * orows) + o) * cpad + c];
    // dw layout (O, I, KH, KW): kyx = ky*KW + kx
    dw[((long)o * Cin + c) * KYX + kyx] = s;
  }
}

// ---------------------------------------------------------------------------

template <int BM>
static void cg_fwd_dispatch(const __bf16* x, const __bf16* x2,
                            const __bf16* wpk, const float* bias, __bf16* out,
                            __bf16* out2, const __bf16* zpage, long Mtot,
                            int HH, int WW, int srcH, int srcW, int sH,
                            int sW, int ld_x, int ld_x2, int C1, int Cin,
                            int Cout, int cpad, int KH, int KW, int offH,
                            int offW, int osplit, int act, int smode,
                            hipStream_t stream) {
  const int tiles_m = (int)((Mtot + BM - 1) / BM);
  const int tiles_n = fh_cdiv(Cout, CG_BN);
  dim3 grid(tiles_m * tiles_n), block(BM == 64 ? 256 : 512);
#define CG_LAUNCH(A, S)                                                      \
  hipLaunchKernelGGL((conv_gemm_fwd_kernel<A, BM, S>), grid, block, 0,       \
                     stream, x, x2, wpk, bias, out, out2, zpage, Mtot, HH,   \
                     WW, srcH, srcW, sH, sW, ld_x, ld_x2, C1, Cin, Cout,     \
                     cpad, KH, KW, offH, offW, osplit, tiles_m)
  if (smode == 0) { if (act == 1) CG_LAUNCH(1, 0); else CG_LAUNCH(0, 0); }
  else if (smode == 1) { if (act == 1) CG_LAUNCH(1, 1); else CG_LAUNCH(0, 1); }
  else { if (act == 1) CG_LAUNCH(1, 2); else CG_LAUNCH(0, 2); }
#undef CG_LAUNCH
}

bool flowhip_conv_halo_fwd_launch(const void* x, const void* x2,
                                  const void* wpk, const float* bias,
                                  void* out, const void* zpage, int N,
                                  int H, int W, int ld_x, int ld_x2, int C1,
                                  int Cin, int Cout, int cpad, int ldo,
                                  int KH, int KW, int act,
                                  hipStream_t stream);

void flowhip_conv_gemm_fwd_launch(const void* x, const void* x2,
                                  const void* wpk, const float* bias,
                                  void* out, void* out2, const void* zpage,
                                  long Mtot, int HH, int WW, int srcH,
                                  int srcW, int sH, int sW, int ld_x,
                                  int ld_x2, int C1, int Cin, int Cout,
                                  int cpad, int KH, int KW, int padH,
                                  int padW, int osplit, int act, int smode,
                                  hipStream_t stream) {
  // stride-1 3x3 / 1x5 / 5x1 (the dominant shapes incl. their
  // backward-data): the halo-staged kernel stages the input once for all
  // KH*KW taps
  if (smode == 0 && out2 == nullptr && padH == KH / 2 && padW == KW / 2 &&
      ((KH == 3 && KW == 3) || (KH == 1 && KW == 5) ||
       (KH == 5 && KW == 1))) {
    const int N = (int)(Mtot / ((long)HH * WW));
    if (flowhip_conv_halo_fwd_launch(x, x2, wpk, bias, out, zpage, N, HH,
                                     WW, ld_x, ld_x2, C1, Cin, Cout, cpad,
                                     Cout, KH, KW, act, stream))
      return;
  }
  // tap offsets: direct conv (smode 0/1) reads sy = oy*sH + ky - padH;
  // strided-transposed (smode 2) taps t = iy + ky + (padH - KH + 1)
  const int offH = smode == 2 ? padH - KH + 1 : -padH;
  const int offW = smode == 2 ? padW - KW + 1 : -padW;
  // Tile choice: BM=128 once the grid still fills the 256-CU chip; the
  // encoder shapes (M >= 64k) take it, the update block (M ~ 21k, 168
  // M-tiles at BM=128) keeps the fill-first 64x64 tile.
  const long t128 = ((Mtot + 127) / 128) * fh_cdiv(Cout, CG_BN);
  if (t128 >= 320)
    cg_fwd_dispatch<128>((const __bf16*)x, (const __bf16*)x2,
                         (const __bf16*)wpk, bias, (__bf16*)out,
                         (__bf16*)out2, (const __bf16*)zpage, Mtot, HH, WW,
                         srcH, srcW, sH, sW, ld_x, ld_x2, C1, Cin, Cout,
                         cpad, KH, KW, offH, offW, osplit, act, smode,
                         stream);
  else
    cg_fwd_dispatch<64>((const __bf16*)x, (const __bf16*)x2,
                        (const __bf16*)wpk, bias, (__bf16*)out, (__bf16*)out2,
                        (const __bf16*)zpage, Mtot, HH, WW, srcH, srcW, sH,
                        sW, ld_x, ld_x2, C1, Cin, Cout, cpad, KH, KW, offH,
                        offW, osplit, act, smode, stream);
}

void flowhip_conv_gemm_wrw_launch(const void* dy, const void* x,
                                  const void* x2, float* partials, float* dw,
                                  float* dbias, const void* zpage, long Mtot,
                                  int HH,
                                  int WW, int srcH, int srcW, int sH, int sW,
                                  int ld_x, int ld_x2, int C1, int Cin,
                                  int Cout, int cpad, int KH, int KW,
                                  int padH, int padW, int nchunk,
                                  hipStream_t stream) {
  const int tiles_o = fh_cdiv(Cout, 64);
  const int tiles_c = fh_cdiv(cpad, 64);
  // bias partials live past the weight partials (caller sized the buffer)
  float* biasp = dbias
      ? partials + (long)nchunk * KH * KW * tiles_o * 64 * cpad
      : nullptr;
  dim3 block(CG_THREADS);
  // (a KW=3 tap-sharing wrw variant was measured SLOWER despite 3x less
  // operand re-read: the 4-image LDS set halves occupancy and the
  // encoder working sets are largely L3-resident, so the re-reads were
  // cheaper than modeled — within-box A/B 41.0 vs 42.65 pairs/s)
  dim3 grid(tiles_o * tiles_c, KH * KW, nchunk);
  hipLaunchKernelGGL(conv_gemm_wrw_kernel, grid, block, 0, stream,
                     (const __bf16*)dy, (const __bf16*)x,
                     (const __bf16*)x2, partials, biasp,
                     (const __bf16*)zpage,
                     Mtot, HH, WW, srcH, srcW, sH, sW, ld_x, ld_x2, C1,
                     Cin, Cout, cpad, KH, KW, padH, padW, tiles_o, tiles_c,
                     nchunk);
  const long total = (long)Cout * Cin * KH * KW;
  long rblocks = (total + CG_THREADS - 1) / CG_THREADS;
  if (rblocks > 4096) rblocks = 4096;
  hipLaunchKernelGGL(conv_gemm_wrw_reduce_kernel, dim3((int)rblocks), block,
                     0, stream, partials, dw, biasp, dbias, nchunk, KH * KW,
                     tiles_o * 64, cpad, Cout, Cin);
}

// ---------------------------------------------------------------------------
// Weight packing (host-side cache refill after every optimizer step): one
// gather kernel instead of the ~5-op torch chain (permute/reshape/pad/cast/
// contiguous) per pack, ~52 packs per training step.
//   fwd pack:  wpk[kyx][o][c]  = c < I ? w[o][c][ky][kx] : 0     (bf16)
//   bwd pack:  wpk[kyx][i][oc] = oc < O ? w[oc][i][KH-1-ky][KW-1-kx] : 0
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256) void conv_gemm_pack_kernel(
    const float* __restrict__ w,  // (O, I, KH, KW)
    __bf16* __restrict__ wpk,     // (KYX, R0, cpad)
    long total, int O, int I, int KH, int KW, int cpad, int flip) {
  const int KYX = KH * KW;
  for (long idx = (long)blockIdx.x * 256 + threadIdx.x; idx < total;
       idx += (long)gridDim.x * 256) {
    long t = idx;
    const int c = t % cpad; t /= cpad;
    const int r = t % (flip ? I : O); t /= (flip ? I : O);
    const int kyx = (int)t;
    float v = 0.f;
    if (!flip) {
      if (c < I) v = w[(((long)r * I + c) * KYX) + kyx];
    } else {
      // r = input channel, c = output channel, spatially flipped
      const int ky = kyx / KW, kx = kyx - (kyx / KW) * KW;
      const int fk = (KH - 1 - ky) * KW + (KW - 1 - kx);
      if (c < O) v = w[(((long)c * I + r) * KYX) + fk];
    }
    wpk[idx] = (__bf16)v;
  }
}

void flowhip_conv_gemm_pack_launch(const float* w, void* wpk, long total,
                                   int O, int I, int KH, int KW, int cpad,
                                   int flip, hipStream_t stream) {
  long blocks = (total + 255) / 256;
  if (blocks > 4096) blocks = 4096;
  hipLaunchKernelGGL(conv_gemm_pack_kernel, dim3((int)blocks), dim3(256), 0,
                     stream, w, (__bf16*)wpk, total, O, I, KH, KW, cpad,
                     flip);
}
