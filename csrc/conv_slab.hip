// Per-image LDS-slab conv kernels (gfx950).
//
// The implicit-GEMM conv kernels in gemm_tile.hip gather their A operand
// from global memory once per (m, k) — the 5x5 im2col expansion re-reads
// every activation element 25x (L3-served, latency-bound: conv_dx measured
// 658us at B=8192).  These kernels instead stage ONE image (+2-pixel halo,
// zero-padded) in LDS and feed the MFMA A-fragments DIRECTLY from the slab
// at shifted offsets — each activation byte crosses the fabric exactly
// once, and the 25 filter taps become LDS address arithmetic.
//
// Geometry (LeNet conv2, H=W=14): BM=224 rows = one image's 196 outputs
// (pool-grouped for fwd / input pixels for dX) + pad; 4 waves as 2x2,
// wave tile 112x(BN/2); one block per image.

#include "common.h"
#include "kernels.h"
#include <stdlib.h>

#define NTHREADS 256
#define BK 64
#define LDK (BK + 8)

typedef ushort_t u16;

__device__ __align__(16) static unsigned short g_zero_page_slab[8];  // glds OOB redirect

// read elements [s, s+8) of an LDS row via two ALIGNED ds_read_b128 + a
// constant shuffle: s is wave-uniform per call site (the 8-case switch is a
// uniform scalar branch), so this replaces 8 scalar ds_read_u16 per
// fragment (the conv-dW x operand is pixel-shifted by the filter tap and
// never 16B-aligned).
DEV short8 lds_row_shifted(const u16* row, int s) {
  int s0 = s & ~7;
  short8 lo = *reinterpret_cast<const short8*>(row + s0);
  short8 hi = *reinterpret_cast<const short8*>(row + s0 + 8);
  switch (s & 7) {
    case 0: return lo;
    case 1: return __builtin_shufflevector(lo, hi, 1, 2, 3, 4, 5, 6, 7, 8);
    case 2: return __builtin_shufflevector(lo, hi, 2, 3, 4, 5, 6, 7, 8, 9);
    case 3: return __builtin_shufflevector(lo, hi, 3, 4, 5, 6, 7, 8, 9, 10);
    case 4: return __builtin_shufflevector(lo, hi, 4, 5, 6, 7, 8, 9, 10, 11);
    case 5: return __builtin_shufflevector(lo, hi, 5, 6, 7, 8, 9, 10, 11, 12);
    case 6: return __builtin_shufflevector(lo, hi, 6, 7, 8, 9, 10, 11, 12, 13);
    default: return __builtin_shufflevector(lo, hi, 7, 8, 9, 10, 11, 12, 13, 14);
  }
}

// ---------------------------------------------------------------------------
// conv2 fwd: y = maxpool(relu(conv5x5(x) + b)), one image per block.
// x slab: [H+4][W+4][Cin] bf16, zero halo. K = 25*Cin, BK = 64.
// ---------------------------------------------------------------------------
template <int H, int W, int CIN, int COUT>
__global__ __launch_bounds__(NTHREADS)
void conv_fwd_slab_kernel(const u16* __restrict__ x,
                          const u16* __restrict__ w,  // [25*CIN][COUT] k-major
                          const float* __restrict__ bias, u16* __restrict__ y,
                          uint8_t* __restrict__ amax, int NB) {
  constexpr int HP = H + 4, WP = W + 4;
  constexpr int HO = H / 2, WO = W / 2;
  constexpr int M = HO * WO * 4;       // pool-grouped outputs of one image
  constexpr int BM = 224;
  constexpr int BN = COUT;             // 64
  constexpr int WM = 112, WN = BN / 2;
  constexpr int MI = WM / 16, NI = WN / 16;
  constexpr int K = 25 * CIN;
  // pixel stride padded 32->40 elems: an A-fragment's 16 lanes read 16
  // DIFFERENT pixels at this stride; 64B stride = 16-way ds_read_b128 bank
  // conflict, 80B = conflict-free ((a/4)%64 steps of 20)
  constexpr int PST = CIN + 8;  // 80B pixel stride: 16B-aligned b128, conflict-reduced
  __shared__ __align__(16) u16 slab[HP * WP * PST];
  // weight tile: K-MAJOR + double-buffered, staged by glds and read with
  // ds_read_b64_tr_b16 (same recipe as dw_tr.hip — no scatter-transpose,
  // the whole next tile is in flight under the current tile's MFMAs)
  __shared__ __align__(16) u16 Bs[2][BK][BN];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int img = blockIdx.x;
  const u16* xi = x + (size_t)img * H * W * CIN;

  constexpr int BLPR = BN / 8;   // lanes per Bs row (8)
  constexpr int BQ = BN / 32;    // glds per wave (2)
  int bnc[BQ];                   // logical n column this lane stages
#pragma unroll
  for (int q = 0; q < BQ; ++q) {
    int rt = (wave * BQ + q) * (512 / BN) + lane / BLPR;
    int xs = (rt ^ (rt >> 3)) & (BLPR - 1);  // XOR bank swizzle (see dw_tr)
    bnc[q] = ((lane % BLPR) ^ xs) * 8;
  }
  const auto issueB = [&](int buf, int kt) {
#pragma unroll
    for (int q = 0; q < BQ; ++q) {
      int row0 = (wave * BQ + q) * (512 / BN);
      int k = kt + row0 + lane / BLPR;
      const u16* src =
          (k < K) ? w + (size_t)k * COUT + bnc[q] : g_zero_page_slab;
      glds16(src, &Bs[buf][row0][0]);
    }
  };

  // stage padded slab (zero halo): chunks of 8 ci
  for (int c = tid; c < HP * WP * (CIN / 8); c += NTHREADS) {
    int ci = (c % (CIN / 8)) * 8;
    int pix = c / (CIN / 8);
    int xx = pix % WP, yy = pix / WP;
    short8 v = short8{0, 0, 0, 0, 0, 0, 0, 0};
    int sy = yy - 2, sx = xx - 2;
    if (sy >= 0 && sy < H && sx >= 0 && sx < W)
      v = *reinterpret_cast<const short8*>(xi + ((size_t)sy * W + sx) * CIN + ci);
    *reinterpret_cast<short8*>(&slab[(yy * WP + xx) * PST + ci]) = v;
  }

  // per-lane A-fragment slab offsets for this wave's MI row-fragments:
  // row lm -> conv-output pixel, pool-grouped (lm = q*4 + pos)
  int arow_off[MI];  // slab element offset of (oy, ox) for each frag row
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
    int lm = wr * WM + mi * 16 + (lane & 15);
    int q = lm >> 2, pos = lm & 3;
    int wo = q % WO, ho = (q / WO) % HO;
    if (lm >= M) { wo = 0; ho = 0; pos = 0; }
    int oy = ho * 2 + (pos >> 1) + 2;   // +2: padded coords
    int ox = wo * 2 + (pos & 1) + 2;
    arow_off[mi] = (oy * WP + ox) * PST;
  }

  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kq = (lane >> 4) * 8;
  const int slot_r = (lane & 15) >> 2;  // tr16 slot addressing (see dw_tr)
  const int slot_c = (lane & 3) * 4;
  const int kgrp = (lane >> 4) * 8;
  issueB(0, 0);    // in flight under the slab-staging barrier's drain
  __syncthreads();

  constexpr int NT_K = (K + BK - 1) / BK;
  int cur = 0;
  for (int t = 0; t < NT_K; ++t) {
    int kt = t * BK;
    if (t + 1 < NT_K) issueB(cur ^ 1, kt + BK);  // hides under MFMAs
    unsigned bbase = (unsigned)(uintptr_t)&Bs[cur][0][0];
#pragma unroll
    for (int kh2 = 0; kh2 < 2; ++kh2) {
      int kbase = kt + kh2 * 32 + kq;       // k = khkw*CIN + ci
      int khkw = kbase / CIN, ci = kbase % CIN;
      // K-tail guard: K=25*CIN is not a multiple of BK, so the last
      // K-step's upper fragments decode khkw >= 25.  Their B operand is
      // zero (staged with a k<K bound) so the product vanishes — but the
      // A read MUST stay inside the slab: an out-of-allocation LDS read
      // can return a NaN bit pattern and NaN*0 = NaN (this was a
      // run-order-dependent test flake).
      if (khkw >= 25) { khkw = 0; ci = 0; }
      int kh = khkw / 5, kw = khkw % 5;
      int shift = ((kh - 2) * WP + (kw - 2)) * PST + ci;
      short8 af[MI], bf[NI];
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
        af[mi] = *reinterpret_cast<const short8*>(
            &slab[arow_off[mi] + shift]);
      uint2 br[NI][2];
#pragma unroll
      for (int ni = 0; ni < NI; ++ni) {
        int cb = (wc * WN + ni * 16 + slot_c) * 2;
        int r0 = kh2 * 32 + kgrp + slot_r;
        int xs0 = (r0 ^ (r0 >> 3)) & (BLPR - 1);
        int r1 = r0 + 4;
        int xs1 = (r1 ^ (r1 >> 3)) & (BLPR - 1);
        br[ni][0] = tr16_issue(bbase + (unsigned)(r0 * (BN * 2) +
                       (((cb >> 4) ^ xs0) << 4) + (cb & 15)));
        br[ni][1] = tr16_issue(bbase + (unsigned)(r1 * (BN * 2) +
                       (((cb >> 4) ^ xs1) << 4) + (cb & 15)));
      }
#pragma unroll
      for (int ni = 0; ni < NI; ++ni) bf[ni] = pack_wait(br[ni][0], br[ni][1]);
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
#pragma unroll
        for (int ni = 0; ni < NI; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();  // drains the in-flight glds (2-buffer overlap)
    cur ^= 1;
  }

  // pool epilogue (per-lane 4-register max, frow multiple of 4)
  const int frow = (lane >> 4) * 4;
  const int fcol = lane & 15;
  u16* yi = y + (size_t)img * (M / 4) * COUT;
  uint8_t* ai = amax + (size_t)img * (M / 4) * COUT;
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
    int lm = wr * WM + mi * 16 + frow;
    if (lm >= M) continue;
    int gq = lm >> 2;
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int gc = wc * WN + ni * 16 + fcol;
      float bias_v = bias[gc];
      float best = -1.0f / 0.0f;
      int barg = 0;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float v = acc[mi][ni][r] + bias_v;
        v = v > 0.f ? v : 0.f;
        if (v > best) { best = v; barg = r; }
      }
      yi[(size_t)gq * COUT + gc] = f2bf(best);
      // liveness rides in the argmax byte: 7 = dead window (relu zeroed
      // all 4) so the pooled-consumer backward kernels need no y read
      ai[(size_t)gq * COUT + gc] = (uint8_t)(best > 0.f ? barg : 7);
    }
  }
}

// ---------------------------------------------------------------------------
// conv dX: dx[hi,wi,ci] = sum_{kh,kw,co} dact[hi-kh+2, wi-kw+2, co] *
//          w[kh,kw,ci,co] — dact slab per image, W read from global per tile.
// ---------------------------------------------------------------------------
template <int H, int W, int CIN, int COUT, int DB = 2, int PAD = 8>
__global__ __launch_bounds__(NTHREADS)
void conv_dx_slab_kernel(const u16* __restrict__ dact,
                         const u16* __restrict__ w,  // [25*CIN][COUT]
                         u16* __restrict__ dx, int NB) {
  constexpr int HP = H + 4, WP = W + 4;
  constexpr int M = H * W;
  constexpr int BM = 224;
  constexpr int BN = CIN;              // 32
  constexpr int WM = 112, WN = BN / 2; // 16
  constexpr int MI = WM / 16, NI = WN / 16;  // 7, 1
  constexpr int K = 25 * COUT;
  // pixel stride: +8 pad = 144B (the original conflict-free choice); +4 =
  // 136B keeps the b128 phases on distinct banks too (34-bank lane step,
  // verified distinct across each 8-lane phase) and shrinks the slab
  // 44.1 vs 46.7 KB — with DB=2's 8 KB of B buffers that is the difference
  // between 2 and 3 resident blocks/CU
  constexpr int PST = COUT + PAD;
  __shared__ __align__(16) u16 slab[HP * WP * PST];
  // weight tile: n-major rows of BK (w rows are already k-contiguous per
  // ci), double-buffered and staged by glds — one barrier per K-step, the
  // next tile in flight under the MFMAs.  The row stride is an unpadded
  // 128B, so the 16B k-chunks are XOR-permuted per row (staging source and
  // fragment read agree on xs(row)) to keep the b128 fragment reads off a
  // single bank pair.
  // DB=2: double-buffered, glds for tile t+1 in flight under tile t's
  // MFMAs, one barrier/K-step — but the extra 4 KB costs a resident block
  // (2/CU) at this slab size.  DB=1: serial glds drain, two barriers, 3
  // blocks/CU — the TLP wins at large NB (same threshold the hi-occ
  // scatter variant sat on).  DB=3: paired K-steps (two buffers staged per
  // barrier pair — half the barriers at DB=1's footprint).
  constexpr int NBUF = (DB >= 2) ? 2 : 1;
  __shared__ __align__(16) u16 Bs[NBUF][BN][BK];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int img = blockIdx.x;
  const u16* di = dact + (size_t)img * H * W * COUT;

  // glds staging decode: one 1KB glds per wave covers rows wave*8..+7
  const int brow = wave * 8 + lane / 8;            // ci row this lane stages
  const int bxs = (brow ^ (brow >> 3)) & 7;
  const int blc = (lane % 8) ^ bxs;                // logical k-chunk
  const auto issueB = [&](int buf, int kt) {
    int k = kt + blc * 8;
    const u16* src = g_zero_page_slab;
    if (k < K) {
      int khkw = k / COUT, co = k % COUT;
      src = w + ((size_t)khkw * CIN + brow) * COUT + co;
    }
    glds16(src, &Bs[buf][wave * 8][0]);
  };

  for (int c = tid; c < HP * WP * (COUT / 8); c += NTHREADS) {
    int co = (c % (COUT / 8)) * 8;
    int pix = c / (COUT / 8);
    int xx = pix % WP, yy = pix / WP;
    short8 v = short8{0, 0, 0, 0, 0, 0, 0, 0};
    int sy = yy - 2, sx = xx - 2;
    if (sy >= 0 && sy < H && sx >= 0 && sx < W)
      v = *reinterpret_cast<const short8*>(di + ((size_t)sy * W + sx) * COUT + co);
    *reinterpret_cast<short8*>(&slab[(yy * WP + xx) * PST + co]) = v;
  }

  int arow_off[MI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
    int lm = wr * WM + mi * 16 + (lane & 15);
    int wi = lm % W, hi = (lm / W) % H;
    if (lm >= M) { wi = 0; hi = 0; }
    arow_off[mi] = ((hi + 2) * WP + (wi + 2)) * PST;  // padded (hi,wi)
  }

  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kq = (lane >> 4) * 8;
  if (DB == 2) issueB(0, 0);  // drains at the slab-staging barrier
  __syncthreads();

  constexpr int NT_K = (K + BK - 1) / BK;
  int cur = 0;
  // DB=3: PAIRED K-steps — both 64-wide buffers staged per barrier pair,
  // halving the barrier count vs DB=1 (13 pairs instead of 25 steps for
  // K=1600) at DB=1's occupancy; the two glds streams drain together.
  if (DB == 3) {
    for (int t = 0; t < NT_K; t += 2) {
      issueB(0, t * BK);
      if (t + 1 < NT_K) issueB(1, (t + 1) * BK);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
#pragma unroll
      for (int half = 0; half < 2; ++half) {
        if (t + half >= NT_K) break;
        int kt = (t + half) * BK;
        const u16* bp = &Bs[half][0][0];
#pragma unroll
        for (int kh2 = 0; kh2 < 2; ++kh2) {
          int kbase = kt + kh2 * 32 + kq;
          int khkw = kbase / COUT, co = kbase % COUT;
          int kh = khkw / 5, kw = khkw % 5;
          int shift = (-(kh - 2) * WP - (kw - 2)) * PST + co;
          short8 af[MI], bf[NI];
#pragma unroll
          for (int mi = 0; mi < MI; ++mi)
            af[mi] = *reinterpret_cast<const short8*>(
                &slab[arow_off[mi] + shift]);
#pragma unroll
          for (int ni = 0; ni < NI; ++ni) {
            int n = wc * WN + ni * 16 + (lane & 15);
            int xs = (n ^ (n >> 3)) & 7;
            int chunk = (kh2 * 4 + (lane >> 4)) ^ xs;
            bf[ni] = *reinterpret_cast<const short8*>(bp + (size_t)n * BK +
                                                      chunk * 8);
          }
#pragma unroll
          for (int mi = 0; mi < MI; ++mi)
#pragma unroll
            for (int ni = 0; ni < NI; ++ni)
              acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
        }
      }
      __syncthreads();
    }
    goto epilogue_dx;
  }
  for (int t = 0; t < NT_K; ++t) {
    int kt = t * BK;
    if (DB == 2) {
      if (t + 1 < NT_K) issueB(cur ^ 1, kt + BK);  // hides under MFMAs
    } else {
      issueB(0, kt);
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __syncthreads();
    }
    const u16* bp = &Bs[cur][0][0];
#pragma unroll
    for (int kh2 = 0; kh2 < 2; ++kh2) {
      int kbase = kt + kh2 * 32 + kq;       // k = khkw*COUT + co
      int khkw = kbase / COUT, co = kbase % COUT;
      int kh = khkw / 5, kw = khkw % 5;
      // dact[hi - kh + 2, wi - kw + 2] -> padded offset shift
      int shift = (-(kh - 2) * WP - (kw - 2)) * PST + co;
      short8 af[MI], bf[NI];
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
        af[mi] = *reinterpret_cast<const short8*>(
            &slab[arow_off[mi] + shift]);
#pragma unroll
      for (int ni = 0; ni < NI; ++ni) {
        int n = wc * WN + ni * 16 + (lane & 15);
        int xs = (n ^ (n >> 3)) & 7;
        int chunk = (kh2 * 4 + (lane >> 4)) ^ xs;
        bf[ni] = *reinterpret_cast<const short8*>(bp + (size_t)n * BK +
                                                  chunk * 8);
      }
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
#pragma unroll
        for (int ni = 0; ni < NI; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();  // DB=2: drains the in-flight glds
    if (DB == 2) cur ^= 1;
  }

epilogue_dx:
  const int frow = (lane >> 4) * 4;
  const int fcol = lane & 15;
  u16* xo = dx + (size_t)img * M * CIN;
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int lm = wr * WM + mi * 16 + frow + r;
      if (lm >= M) continue;
#pragma unroll
      for (int ni = 0; ni < NI; ++ni) {
        int gc = wc * WN + ni * 16 + fcol;
        xo[(size_t)lm * CIN + gc] = f2bf(acc[mi][ni][r]);
      }
    }
  }
}

// High-occupancy variant of conv dX (the pre-glds staging: padded
// n-major Bs, 2 barriers/K-step, 3 blocks/CU).  The glds-pipelined
// kernel above is faster at small NB but its extra LDS buffer drops
// occupancy to 2 blocks/CU, which loses at NB >= 2048 (measured
// 371 vs 327 us at B=8192); the launcher picks by NB.
template <int H, int W, int CIN, int COUT>
__global__ __launch_bounds__(NTHREADS)
void conv_dx_slab_hi_occ_kernel(const u16* __restrict__ dact,
                         const u16* __restrict__ w,  // [25*CIN][COUT]
                         u16* __restrict__ dx, int NB) {
  constexpr int HP = H + 4, WP = W + 4;
  constexpr int M = H * W;
  constexpr int BM = 224;
  constexpr int BN = CIN;              // 32
  constexpr int WM = 112, WN = BN / 2; // 16
  constexpr int MI = WM / 16, NI = WN / 16;  // 7, 1
  constexpr int K = 25 * COUT;
  constexpr int PST = COUT + 8;  // 144B pixel stride: conflict-free b128
  __shared__ __align__(16) u16 slab[HP * WP * PST];
  __shared__ __align__(16) u16 Bs[BN][LDK];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  const int img = blockIdx.x;
  const u16* di = dact + (size_t)img * H * W * COUT;

  for (int c = tid; c < HP * WP * (COUT / 8); c += NTHREADS) {
    int co = (c % (COUT / 8)) * 8;
    int pix = c / (COUT / 8);
    int xx = pix % WP, yy = pix / WP;
    short8 v = short8{0, 0, 0, 0, 0, 0, 0, 0};
    int sy = yy - 2, sx = xx - 2;
    if (sy >= 0 && sy < H && sx >= 0 && sx < W)
      v = *reinterpret_cast<const short8*>(di + ((size_t)sy * W + sx) * COUT + co);
    *reinterpret_cast<short8*>(&slab[(yy * WP + xx) * PST + co]) = v;
  }

  int arow_off[MI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
    int lm = wr * WM + mi * 16 + (lane & 15);
    int wi = lm % W, hi = (lm / W) % H;
    if (lm >= M) { wi = 0; hi = 0; }
    arow_off[mi] = ((hi + 2) * WP + (wi + 2)) * PST;  // padded (hi,wi)
  }

  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kq = (lane >> 4) * 8;
  __syncthreads();

  for (int kt = 0; kt < K; kt += BK) {
    // Bs[ci][kk] = w[((khkw)*CIN + ci)*COUT + co(k)] — rows16 from w slices
    for (int c = tid; c < BN * (BK / 8); c += NTHREADS) {
      int i = c / (BK / 8);
      int kc = (c % (BK / 8)) * 8;
      int k = kt + kc;
      int khkw = k / COUT, co = k % COUT;
      short8 v = *reinterpret_cast<const short8*>(
          w + ((size_t)khkw * CIN + i) * COUT + co);
      *reinterpret_cast<short8*>(&Bs[i][kc]) = v;
    }
    __syncthreads();
#pragma unroll
    for (int kh2 = 0; kh2 < 2; ++kh2) {
      int kbase = kt + kh2 * 32 + kq;       // k = khkw*COUT + co
      int khkw = kbase / COUT, co = kbase % COUT;
      int kh = khkw / 5, kw = khkw % 5;
      // dact[hi - kh + 2, wi - kw + 2] -> padded offset shift
      int shift = (-(kh - 2) * WP - (kw - 2)) * PST + co;
      short8 af[MI], bf[NI];
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
        af[mi] = *reinterpret_cast<const short8*>(
            &slab[arow_off[mi] + shift]);
#pragma unroll
      for (int ni = 0; ni < NI; ++ni)
        bf[ni] = *reinterpret_cast<const short8*>(
            &Bs[wc * WN + ni * 16 + (lane & 15)][kh2 * 32 + kq]);
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
#pragma unroll
        for (int ni = 0; ni < NI; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    __syncthreads();
  }

  const int frow = (lane >> 4) * 4;
  const int fcol = lane & 15;
  u16* xo = dx + (size_t)img * M * CIN;
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int lm = wr * WM + mi * 16 + frow + r;
      if (lm >= M) continue;
#pragma unroll
      for (int ni = 0; ni < NI; ++ni) {
        int gc = wc * WN + ni * 16 + fcol;
        xo[(size_t)lm * CIN + gc] = f2bf(acc[mi][ni][r]);
      }
    }
  }
}

// ---- host wrappers --------------------------------------------------------
void launch_conv_fwd_slab(const unsigned short* x, const unsigned short* w,
                          const float* bias, unsigned short* y, uint8_t* amax,
                          int NB, int H, int W, int Cin, int Cout,
                          hipStream_t s) {
  if (H == 14 && W == 14 && Cin == 32 && Cout == 64) {
    hipLaunchKernelGGL((conv_fwd_slab_kernel<14, 14, 32, 64>), dim3(NB),
                       dim3(NTHREADS), 0, s, x, w, bias, y, amax, NB);
  } else {
    // unsupported geometry falls back at the binding level
  }
}

bool conv_slab_supported(int H, int W, int Cin, int Cout) {
  return H == 14 && W == 14 && Cin == 32 && Cout == 64;
}



void launch_conv_dx_slab(const unsigned short* dact, const unsigned short* w,
                         unsigned short* dx, int NB, int H, int W, int Cin,
                         int Cout, hipStream_t s) {
  if (H == 14 && W == 14 && Cin == 32 && Cout == 64) {
    // glds-pipelined (double-buffered) Bs wins below NB=2048; its extra
    // LDS buffer costs a resident block (3 -> 2 per CU), which TLP-bound
    // large batches feel — there the SINGLE-buffered glds form keeps
    // 3/CU.  DMNIST_DX_HIOCC=1 reverts large batches to the original
    // scatter-staged kernel for A/B.
    if (getenv("DMNIST_DX_PAIR"))  // paired-K ablation (half the barriers)
      hipLaunchKernelGGL((conv_dx_slab_kernel<14, 14, 32, 64, 3>), dim3(NB),
                         dim3(NTHREADS), 0, s, dact, w, dx, NB);
    else if (NB < 2048)
      hipLaunchKernelGGL((conv_dx_slab_kernel<14, 14, 32, 64, 2>), dim3(NB),
                         dim3(NTHREADS), 0, s, dact, w, dx, NB);
    else if (getenv("DMNIST_DX_HIOCC"))
      hipLaunchKernelGGL((conv_dx_slab_hi_occ_kernel<14, 14, 32, 64>),
                         dim3(NB), dim3(NTHREADS), 0, s, dact, w, dx, NB);
    else if (getenv("DMNIST_DX_DB2SLIM"))
      // glds-pipelined DB=2 at the slim PAD=4 slab (3 blocks/CU) —
      // measured SLOWER than DB=1 at 8192 (357 vs 320 us solo) despite the
      // occupancy win; kept for A/B
      hipLaunchKernelGGL((conv_dx_slab_kernel<14, 14, 32, 64, 2, 4>),
                         dim3(NB), dim3(NTHREADS), 0, s, dact, w, dx, NB);
    else
      hipLaunchKernelGGL((conv_dx_slab_kernel<14, 14, 32, 64, 1>), dim3(NB),
                         dim3(NTHREADS), 0, s, dact, w, dx, NB);
  }
}

// ---------------------------------------------------------------------------
// conv dW (Cin=32, Cout=64): dW[kh,kw,ci,co] = sum_{n,h,w} x[n,h+kh-2,w+kw-2,ci]
//   * dact[n,h,w,co], computed TRANSPOSED (M'=co=64, N'=(khkw,ci)=800) so the
//   A operand (dact^T slab, pixel-contiguous per co row) reads as aligned
//   ds_read_b128 fragments.
//
// Padded-linear trick: the reduction k enumerates the PADDED 18x18 pixel
// space (halo rows/cols of dact^T are zero, so halo k contribute nothing).
// The x operand address is then LINEAR in k: a = (kh*18 + kw) + k, over an
// x slab stored per-ci with implicit row stride 18 covering rows -4..17 and
// cols -4..13 (entry (r,c) = x[r-4][c-4], zero outside).  A column overflow
// wp+kw >= 18 wraps to (row+1, col-18) whose col-4 < 0 is halo-ZERO — which
// equals the true out-of-range-zero value, so the wrap is CORRECT by
// construction (no per-element bounds checks on the hot path).
//
// Block: G=16 images accumulated into the same registers (flush atomics
// /16); 2 N'-tiles of 448 (waves 1x4, wave tile 64x112, acc 4x7).
// ---------------------------------------------------------------------------
// ABL: perf-ablation variants (0=full, 1=skip MFMA loop, 2=skip staging,
// 3=skip flush) — selected by DMNIST_DW_ABL, numerically wrong except 0.
template <int H, int W, int CIN, int COUT, int G, int ABL = 0>
__global__ __launch_bounds__(NTHREADS, 2)  // force <=256 regs: 2 blocks/CU
void conv_dw_slab_kernel(const u16* __restrict__ x,
                         const u16* __restrict__ dact,
                         float* __restrict__ dw, int NB) {
  constexpr int HP = H + 4, WP = W + 4;           // 18 x 18
  constexpr int KPAD = 384;                       // 324 padded to 6 BK-steps
  constexpr int DST = KPAD + 8;                   // row stride: the 16 lanes
  // of an A-fragment read 16 co ROWS; a 768B stride is a 16-way b128 bank
  // conflict, 784B is conflict-free
  constexpr int XROW = 472;                       // max addr 466 + margin
  constexpr int NP = 25 * CIN;                    // 800
  constexpr int BN = 448;                         // 2 tiles
  constexpr int WN = 112, NI = WN / 16;           // 7
  constexpr int MI = COUT / 16;                   // 4 (WM = COUT = 64)
  __shared__ __align__(16) u16 dslab[COUT][DST];
  __shared__ __align__(16) u16 xslab[CIN][XROW];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wc = tid >> 6;  // wave = N position (waves 1x4)
  const int img0 = blockIdx.x * G;

  const int lcol = lane & 15;
  const int kq = (lane >> 4) * 8;

  for (int nt = 0; nt < 2; ++nt) {
    f32x4 acc[MI][NI];
#pragma unroll
    for (int mi = 0; mi < MI; ++mi)
#pragma unroll
      for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

    // per-lane B (x slab) base offsets for this tile's columns
    int bshift[NI];
    int bci[NI];
    bool bvalid[NI];
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int np = nt * BN + wc * WN + ni * 16 + lcol;
      int khkw = np / CIN, ci = np % CIN;
      bvalid[ni] = np < NP;
      if (!bvalid[ni]) { khkw = 0; ci = 0; }
      bci[ni] = ci;
      bshift[ni] = (khkw / 5) * WP + (khkw % 5);
    }

    // zero-fill ONCE per tile pass: the halo/pad zeros survive across
    // images (staging overwrites exactly the valid pixel region each time)
    __syncthreads();
    for (int c = tid; c < COUT * (DST / 8); c += NTHREADS) {
      *reinterpret_cast<short8*>(&dslab[c / (DST / 8)][(c % (DST / 8)) * 8]) =
          short8{0, 0, 0, 0, 0, 0, 0, 0};
    }
    for (int c = tid; c < CIN * (XROW / 8); c += NTHREADS) {
      *reinterpret_cast<short8*>(&xslab[c / (XROW / 8)][(c % (XROW / 8)) * 8]) =
          short8{0, 0, 0, 0, 0, 0, 0, 0};
    }
    for (int g = 0; g < G; ++g) {
      int img = img0 + g;
      if (img >= NB) break;
      __syncthreads();  // previous compute / zero pass done before overwrite
      const u16* di = dact + (size_t)img * H * W * COUT;
      const u16* xi = x + (size_t)img * H * W * CIN;
      if (ABL == 2) goto compute;  // slab holds zeros/stale: staging ablated
      for (int c = tid; c < H * W * (COUT / 8); c += NTHREADS) {
        int co0 = (c % (COUT / 8)) * 8;
        int pix = c / (COUT / 8);
        int w_ = pix % W, h_ = pix / W;
        short8 v = *reinterpret_cast<const short8*>(di + (size_t)pix * COUT + co0);
        int pk = (h_ + 2) * WP + (w_ + 2);
#pragma unroll
        for (int e = 0; e < 8; ++e) dslab[co0 + e][pk] = v[e];
      }
      for (int c = tid; c < H * W * (CIN / 8); c += NTHREADS) {
        int ci0 = (c % (CIN / 8)) * 8;
        int pix = c / (CIN / 8);
        int w_ = pix % W, h_ = pix / W;
        short8 v = *reinterpret_cast<const short8*>(xi + (size_t)pix * CIN + ci0);
        int pk = (h_ + 4) * WP + (w_ + 4);
#pragma unroll
        for (int e = 0; e < 8; ++e) xslab[ci0 + e][pk] = v[e];
      }
compute:
      __syncthreads();
      if (ABL == 1) continue;
      // ---- 6 K-steps over the padded pixel space ----
      for (int kt = 0; kt < KPAD; kt += BK) {
#pragma unroll
        for (int kh2 = 0; kh2 < 2; ++kh2) {
          int k0 = kt + kh2 * 32 + kq;
          short8 af[MI];
#pragma unroll
          for (int mi = 0; mi < MI; ++mi)
            af[mi] = *reinterpret_cast<const short8*>(
                &dslab[mi * 16 + lcol][k0]);
          // B fragments just-in-time; bshift is UNIFORM per fragment
          // (a 16-column block never straddles a khkw boundary), so the
          // shifted read is two aligned b128 + a uniform constant shuffle
#pragma unroll
          for (int ni = 0; ni < NI; ++ni) {
            short8 bf = lds_row_shifted(&xslab[bci[ni]][0],
                                        bshift[ni] + k0);
#pragma unroll
            for (int mi = 0; mi < MI; ++mi)
              acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                  af[mi], bf, acc[mi][ni], 0, 0, 0);
          }
        }
      }
    }
    // ---- flush: dW'[co][n'] -> atomicAdd dW[(khkw*CIN+ci)*COUT + co] ----
    if (ABL == 3) { __syncthreads(); continue; }
#pragma unroll
    for (int mi = 0; mi < MI; ++mi) {
#pragma unroll
      for (int ni = 0; ni < NI; ++ni) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int co = mi * 16 + (lane >> 4) * 4 + r;   // C/D row map
          int np = nt * BN + wc * WN + ni * 16 + lcol;
          if (np >= NP) continue;
          int khkw = np / CIN, ci = np % CIN;
          float v = acc[mi][ni][r];
          if (v != 0.f)
            atomicAdd(dw + ((size_t)khkw * CIN + ci) * COUT + co, v);
        }
      }
    }
    __syncthreads();
  }
}

void launch_conv_dw_slab(const unsigned short* x, const unsigned short* dact,
                         float* dw, int NB, int H, int W, int Cin, int Cout,
                         hipStream_t s) {
  if (H == 14 && W == 14 && Cin == 32 && Cout == 64) {
    // pick the group size so the grid has >= ~512 blocks (256 CUs need
    // >>256 workgroups); bigger G amortizes the flush atomics.
    // DMNIST_DW_G overrides for testing the G>1 paths at small NB.
    static int force_g = [] {
      const char* e = getenv("DMNIST_DW_G");
      return e ? atoi(e) : 0;
    }();
    int nb_eff = force_g ? (force_g >= 16 ? 8192 : (force_g >= 4 ? 2048 : 0))
                         : NB;
    static int abl = [] {
      const char* e = getenv("DMNIST_DW_ABL");
      return e ? atoi(e) : 0;
    }();
    if (nb_eff >= 8192) {
      dim3 grid((NB + 15) / 16);
      switch (abl) {
        case 1: hipLaunchKernelGGL((conv_dw_slab_kernel<14, 14, 32, 64, 16, 1>), grid, dim3(NTHREADS), 0, s, x, dact, dw, NB); break;
        case 2: hipLaunchKernelGGL((conv_dw_slab_kernel<14, 14, 32, 64, 16, 2>), grid, dim3(NTHREADS), 0, s, x, dact, dw, NB); break;
        case 3: hipLaunchKernelGGL((conv_dw_slab_kernel<14, 14, 32, 64, 16, 3>), grid, dim3(NTHREADS), 0, s, x, dact, dw, NB); break;
        default: hipLaunchKernelGGL((conv_dw_slab_kernel<14, 14, 32, 64, 16, 0>), grid, dim3(NTHREADS), 0, s, x, dact, dw, NB); break;
      }
    } else if (nb_eff >= 2048) {
      hipLaunchKernelGGL((conv_dw_slab_kernel<14, 14, 32, 64, 4>),
                         dim3((NB + 3) / 4), dim3(NTHREADS), 0, s, x, dact,
                         dw, NB);
    } else {
      hipLaunchKernelGGL((conv_dw_slab_kernel<14, 14, 32, 64, 1>),
                         dim3(NB), dim3(NTHREADS), 0, s, x, dact, dw, NB);
    }
  }
}

// ---------------------------------------------------------------------------
// conv1 dW (Cin=1): same padded-linear transposed formulation as above but
// with the trivial N' = 25 taps (no ci): waves 2x2, one 16x16 fragment per
// wave covers the whole 32(co) x 32(khkw-padded) output.
// ---------------------------------------------------------------------------
template <int H, int W, int COUT, int G>
__global__ __launch_bounds__(NTHREADS, 2)
void conv1_dw_slab_kernel(const u16* __restrict__ x,
                          const u16* __restrict__ dact,
                          float* __restrict__ dw, int NB) {
  constexpr int HP = H + 4, WP = W + 4;            // 32 x 32
  constexpr int KPAD = HP * WP;                    // 1024 = 16 BK-steps
  constexpr int DST = KPAD + 8;  // 2064B row stride: conflict-free b128
  constexpr int XROW = 1168;  // max addr (4*32+4) + 1023 + 8 margin
  __shared__ __align__(16) u16 dslab[COUT][DST];
  __shared__ __align__(16) u16 xslab[XROW];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;         // 2x2 over [32co][32khkw]
  const int img0 = blockIdx.x * G;
  const int lcol = lane & 15;
  const int kq = (lane >> 4) * 8;

  const int khkw = wc * 16 + lcol;
  const int kh = khkw / 5, kw = khkw % 5;
  const int bshift = (khkw < 25) ? kh * WP + kw : 0;

  f32x4 acc = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int c = tid; c < COUT * (DST / 8); c += NTHREADS)
    *reinterpret_cast<short8*>(&dslab[c / (DST / 8)][(c % (DST / 8)) * 8]) =
        short8{0, 0, 0, 0, 0, 0, 0, 0};
  for (int c = tid; c < XROW / 8; c += NTHREADS)
    *reinterpret_cast<short8*>(&xslab[c * 8]) = short8{0, 0, 0, 0, 0, 0, 0, 0};
  for (int g = 0; g < G; ++g) {
    int img = img0 + g;
    if (img >= NB) break;
    __syncthreads();  // previous compute / zero pass done before overwrite
    const u16* di = dact + (size_t)img * H * W * COUT;
    const u16* xi = x + (size_t)img * H * W;
    for (int c = tid; c < H * W * (COUT / 8); c += NTHREADS) {
      int co0 = (c % (COUT / 8)) * 8;
      int pix = c / (COUT / 8);
      int w_ = pix % W, h_ = pix / W;
      short8 v = *reinterpret_cast<const short8*>(di + (size_t)pix * COUT + co0);
      int pk = (h_ + 2) * WP + (w_ + 2);
#pragma unroll
      for (int e = 0; e < 8; ++e) dslab[co0 + e][pk] = v[e];
    }
    for (int c = tid; c < H * W; c += NTHREADS) {
      int w_ = c % W, h_ = c / W;
      xslab[(h_ + 4) * WP + (w_ + 4)] = xi[c];
    }
    __syncthreads();
    for (int kt = 0; kt < KPAD; kt += BK) {
#pragma unroll
      for (int kh2 = 0; kh2 < 2; ++kh2) {
        int k0 = kt + kh2 * 32 + kq;
        short8 af = *reinterpret_cast<const short8*>(
            &dslab[wr * 16 + lcol][k0]);
        const u16* base = &xslab[bshift + k0];
        short8 bf;
#pragma unroll
        for (int e = 0; e < 8; ++e) bf[e] = (short)base[e];
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
      }
    }
  }
  // flush: D[co][khkw]
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int co = wr * 16 + (lane >> 4) * 4 + r;
    if (khkw < 25) {
      float v = acc[r];
      if (v != 0.f) atomicAdd(dw + (size_t)khkw * COUT + co, v);
    }
  }
}

void launch_conv1_dw_slab(const unsigned short* x, const unsigned short* dact,
                          float* dw, int NB, int H, int W, int Cout,
                          hipStream_t s) {
  if (H == 28 && W == 28 && Cout == 32) {
    if (NB >= 4096) {
      hipLaunchKernelGGL((conv1_dw_slab_kernel<28, 28, 32, 8>),
                         dim3((NB + 7) / 8), dim3(NTHREADS), 0, s, x, dact,
                         dw, NB);
    } else {
      hipLaunchKernelGGL((conv1_dw_slab_kernel<28, 28, 32, 2>),
                         dim3((NB + 1) / 2), dim3(NTHREADS), 0, s, x, dact,
                         dw, NB);
    }
  }
}

bool conv1_slab_supported(int H, int W, int Cin, int Cout) {
  return H == 28 && W == 28 && Cin == 1 && Cout == 32;
}
