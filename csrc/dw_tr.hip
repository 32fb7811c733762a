// tr16 + glds dW GEMM for gfx950 — the "designed around glds" staging the
// guide prescribes for k-strided operands (cdna_hip_programming.md: glds
// pipelining + ds_read_b64_tr_b16; semantics pinned empirically in
// profiles/r01_tr16_probe.md with tools/probes/tr16_probe.hip).
//
// dW = A^T-ish reductions over the batch-pixel axis K: both operands are
// K-MAJOR in memory (x / h2 rows are per-pixel, dact / dyeff rows are
// per-pixel).  The old path (gemm_tile.hip A_T/A_CONV_DW) had to SCATTER
// each staged row element-by-element into an m-major LDS image so the MFMA
// fragment reads were contiguous — 8 ds_write_b16 per 16B of operand, and
// the ds_write's vmcnt wait serialized the global loads (~2 TB/s MLP wall).
// Here the LDS image stays K-MAJOR and LINEAR:
//   * staging is pure glds (global_load_lds_dwordx4): no VGPR round-trip,
//     no scatter, fire-and-forget — a whole tile is in flight while the
//     previous tile's MFMAs run (2-buffer overlap, one barrier/K-step);
//   * fragment reads use ds_read_b64_tr_b16, the LDS transpose read: within
//     each 16-lane group, lane g supplies &img[kq + (g>>2)][n0 + 4*(g&3)]
//     and lane i receives img[kq + j][n0 + i] (j = 0..3) — exactly an MFMA
//     operand fragment, with ZERO staging-time transpose work.
// Out-of-bounds / K-tail lanes redirect their glds source to a zeroed
// device page (glds cannot conditionally mask, and every lane of the wave
// must participate).
//
// Modes: AM_PLAIN  A[m][k] = Asrc[k*lda + m]                (fc dW: x^T)
//        AM_CONV5  A[m=(khkw,ci)][k=pixel] 5x5-SAME gather  (conv dW)
// Requirements: M % 8 == 0, N % BN == 0, Cin % 8 == 0 (conv), 16B-aligned
// sources.  Output: fp32 split-K atomics straight into the grad bucket.

#include "common.h"
#include "kernels.h"

#define NT 256

__device__ __align__(16) unsigned short g_zero_page[8];  // zero-init

enum { AM_PLAIN = 0, AM_CONV5 = 1 };

template <int BN, int AMODE, int TBM = 128, int TBK = 64, int NBUF = 2>
__global__ __launch_bounds__(NT) void dw_tr_kernel(GemmParams p) {
  static_assert(TBK % 32 == 0, "TBK in 32-k MFMA halves");
  static_assert(NBUF == 2 || NBUF == 3, "2-buf overlap or 3-buf glds span");
  constexpr int WM = TBM / 2, WN = BN / 2;
  constexpr int MI = WM / 16, NI = WN / 16;
  constexpr int AQ = TBM * TBK / 2048;  // A glds per wave (1KB each)
  constexpr int BQ = BN * TBK / 2048;   // B glds per wave
  constexpr int ALPR = TBM / 8;       // lanes per A row
  constexpr int BLPR = BN / 8;        // lanes per B row
  __shared__ __align__(16) ushort_t Aimg[NBUF][TBK][TBM];  // k-major, LINEAR
  __shared__ __align__(16) ushort_t Bimg[NBUF][TBK][BN];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;

  // XCD-chunk swizzle on the 1-D grid (same remap as gemm_tile SWZ): blocks
  // sharing a split-K slice land on one XCD so the operand slice stays in
  // that XCD's L2.
  int gx = (p.M + TBM - 1) / TBM;
  int gy = (p.N + BN - 1) / BN;
  int total = gx * gy * p.splitk;
  int flat = blockIdx.x;
  int q8 = total / 8, r8 = total % 8;
  int xcd = flat % 8, pos = flat / 8;
  int flat2 = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + pos;
  const int m0 = (flat2 % gx) * TBM;
  int rest = flat2 / gx;
  const int n0 = (rest % gy) * BN;
  const int bz = rest / gy;

  int ksteps_total = (p.K + TBK - 1) / TBK;
  int steps_per = (ksteps_total + p.splitk - 1) / p.splitk;
  int kbeg = bz * steps_per * TBK;
  int kend = min(p.K, kbeg + steps_per * TBK);
  int nt = (kend - kbeg + TBK - 1) / TBK;
  if (nt < 0) nt = 0;

  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  // ---- per-lane glds source decode (loop-invariant parts) ----------------
  // A: glds gi = wave*AQ+q stages rows gi*(512/TBM) + lane/ALPR at columns
  //    (lane%ALPR)*8..+7; LDS dest = &Aimg[buf][row0][0] + lane*16 (linear).
  //
  // XOR bank swizzle: un-swizzled, every LDS row is a 256B/128B stride =
  // a whole bank wrap, so the tr16 slot reads (4 rows x same columns) were
  // ~8-way bank-conflicted (measured conf/idx = 0.85, profiles/).  The LDS
  // DEST of glds is fixed (base + lane*16), so the swizzle is applied to
  // the SOURCE: the lane staging physical 16B-chunk pc of tile-row rt
  // fetches LOGICAL chunk pc ^ xs(rt), with xs(rt) = (rt ^ (rt>>4)) & 15
  // for A (16 chunks/row) and (rt ^ (rt>>3)) & 7 for B.  rt depends only
  // on (wave, q, lane) — never on the K-tile — so decodes stay loop-
  // invariant.  The tr16 addresses apply the same xs per slot row.
  int amc[AQ];     // logical m column this lane stages, per glds instr
  bool am_ok[AQ];
  int akh[AQ], akw[AQ], aci[AQ];
#pragma unroll
  for (int q = 0; q < AQ; ++q) {
    int rt = (wave * AQ + q) * (512 / TBM) + lane / ALPR;
    int xs = (rt ^ (rt >> 4)) & (ALPR - 1);
    amc[q] = m0 + ((lane % ALPR) ^ xs) * 8;
    am_ok[q] = amc[q] < p.M;
    if (AMODE == AM_CONV5) {
      int khkw = amc[q] / p.Cin;
      aci[q] = amc[q] % p.Cin;
      akh[q] = khkw / 5 - 2;
      akw[q] = khkw % 5 - 2;
    } else {
      akh[q] = akw[q] = aci[q] = 0;
    }
  }
  int bnc[BQ];
  bool bn_ok[BQ];
#pragma unroll
  for (int q = 0; q < BQ; ++q) {
    int rt = (wave * BQ + q) * (512 / BN) + lane / BLPR;
    int xs = (rt ^ (rt >> 3)) & (BLPR - 1);
    bnc[q] = n0 + ((lane % BLPR) ^ xs) * 8;
    bn_ok[q] = bnc[q] < p.N;
  }

  // incremental pixel decode (AM_CONV5): the k this (lane, q) stages
  // advances by exactly TBK per issue() call, so a mixed-radix add with
  // carries replaces the per-tile runtime div/mod chain
  int aw[AQ], ah[AQ], an[AQ];
  int dw64 = 0, dh64 = 0, dn64 = 0;
  if (AMODE == AM_CONV5) {
    dw64 = TBK % p.CW;
    int t64 = TBK / p.CW;
    dh64 = t64 % p.CH;
    dn64 = t64 / p.CH;
#pragma unroll
    for (int q = 0; q < AQ; ++q) {
      int k0 = kbeg + (wave * AQ + q) * (512 / TBM) + lane / ALPR;
      aw[q] = k0 % p.CW;
      int t2 = k0 / p.CW;
      ah[q] = t2 % p.CH;
      an[q] = t2 / p.CH;
    }
  }

  const auto issue = [&](int buf, int kt) {
#pragma unroll
    for (int q = 0; q < AQ; ++q) {
      int gi = wave * AQ + q;
      int row0 = gi * (512 / TBM);
      int k = kt + row0 + lane / ALPR;
      const ushort_t* src = g_zero_page;
      if (k < kend && am_ok[q]) {
        if (AMODE == AM_PLAIN) {
          src = p.A + (size_t)k * p.lda + amc[q];
        } else {
          int y = ah[q] + akh[q], x = aw[q] + akw[q];
          if (y >= 0 && y < p.CH && x >= 0 && x < p.CW)
            src = p.A + (((size_t)an[q] * p.CH + y) * p.CW + x) * p.Cin + aci[q];
        }
      }
      glds16(src, &Aimg[buf][row0][0]);
      if (AMODE == AM_CONV5) {  // advance to the pixel this lane stages next
        aw[q] += dw64;
        if (aw[q] >= p.CW) { aw[q] -= p.CW; ah[q] += 1; }
        ah[q] += dh64;
        if (ah[q] >= p.CH) { ah[q] -= p.CH; an[q] += 1; }
        an[q] += dn64;
      }
    }
#pragma unroll
    for (int q = 0; q < BQ; ++q) {
      int gi = wave * BQ + q;
      int row0 = gi * (512 / BN);
      int k = kt + row0 + lane / BLPR;
      const ushort_t* src = (k < kend && bn_ok[q])
                                ? p.B + (size_t)k * p.ldb + bnc[q]
                                : g_zero_page;
      glds16(src, &Bimg[buf][row0][0]);
    }
  };

  // ---- tr16 fragment addressing (loop-invariant parts) -------------------
  // within each 16-lane group: lane g addresses (row kq + (g>>2), col
  // frag_n0 + 4*(g&3)); the second read is 4 rows further down.
  const int slot_r = (lane & 15) >> 2;
  const int slot_c = (lane & 3) * 4;
  const int kgrp = (lane >> 4) * 8;  // this group's k-offset within the half

  const auto compute = [&](int buf) {
    unsigned abase = (unsigned)(uintptr_t)&Aimg[buf][0][0];
    unsigned bbase = (unsigned)(uintptr_t)&Bimg[buf][0][0];
#pragma unroll
    for (int kh2 = 0; kh2 < TBK / 32; ++kh2) {
      int krow = kh2 * 32 + kgrp + slot_r;
      short8 af[MI], bf[NI];
      // slot address with the matching XOR chunk swizzle (see staging note)
      const auto aswz = [&](int r, int cb) {
        int xs = (r ^ (r >> 4)) & (ALPR - 1);
        return abase + (unsigned)(r * (TBM * 2) + (((cb >> 4) ^ xs) << 4) +
                                  (cb & 15));
      };
      const auto bswz = [&](int r, int cb) {
        int xs = (r ^ (r >> 3)) & (BLPR - 1);
        return bbase + (unsigned)(r * (BN * 2) + (((cb >> 4) ^ xs) << 4) +
                                  (cb & 15));
      };
      uint2 ar[MI][2], br[NI][2];
#pragma unroll
      for (int mi = 0; mi < MI; ++mi) {
        int cb = (wr * WM + mi * 16 + slot_c) * 2;
        ar[mi][0] = tr16_issue(aswz(krow, cb));
        ar[mi][1] = tr16_issue(aswz(krow + 4, cb));
      }
#pragma unroll
      for (int ni = 0; ni < NI; ++ni) {
        int cb = (wc * WN + ni * 16 + slot_c) * 2;
        br[ni][0] = tr16_issue(bswz(krow, cb));
        br[ni][1] = tr16_issue(bswz(krow + 4, cb));
      }
#pragma unroll
      for (int mi = 0; mi < MI; ++mi) af[mi] = pack_wait(ar[mi][0], ar[mi][1]);
#pragma unroll
      for (int ni = 0; ni < NI; ++ni) bf[ni] = pack_wait(br[ni][0], br[ni][1]);
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
#pragma unroll
        for (int ni = 0; ni < NI; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
  };

  constexpr int INFLIGHT = AQ + BQ;  // glds per wave per tile
  if (NBUF == 3) {
    // 3-buffer glds SPAN (guide: counted vmcnt + raw s_barrier): two tiles
    // stay in flight across each barrier; each wave waits only until the
    // tile it is about to READ is complete (its own glds are program-order,
    // so vmcnt(INFLIGHT) leaves exactly the newest tile outstanding).
    if (nt > 0) issue(0, kbeg);
    if (nt > 1) {
      issue(1 % NBUF, kbeg + TBK);
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(INFLIGHT) : "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    for (int t = 0; t < nt; ++t) {
      if (t + 2 < nt) issue((t + 2) % NBUF, kbeg + (t + 2) * TBK);
      compute(t % NBUF);
      if (t + 2 < nt)
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(INFLIGHT) : "memory");
      else
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      __builtin_amdgcn_s_barrier();
    }
  } else {
    int cur = 0;
    if (nt > 0) issue(0, kbeg);
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __syncthreads();
    for (int t = 0; t < nt; ++t) {
      if (t + 1 < nt) issue(cur ^ 1, kbeg + (t + 1) * TBK);  // under MFMAs
      compute(cur);
      // one barrier per K-step; the implicit vmcnt(0) drain at
      // __syncthreads completes tile t+1's glds (2-buffer overlap)
      __syncthreads();
      cur ^= 1;
    }
  }

  const int frow = (lane >> 4) * 4;
  const int fcol = lane & 15;
  float* outp = reinterpret_cast<float*>(p.C);
  // p.offset != 0: DIAGNOSTIC contention probe — spread each k-slice's
  // atomics over 16 separate planes (outputs meaningless; isolates the
  // same-address atomic serialization cost in the flush)
  if (p.offset) outp += (size_t)(bz & 15) * p.M * p.ldc;
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int gc = n0 + wc * WN + ni * 16 + fcol;
      if (gc >= p.N) continue;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int gr = m0 + wr * WM + mi * 16 + frow + r;
        if (gr >= p.M) continue;
        atomicAdd(outp + (size_t)gr * p.ldc + gc, acc[mi][ni][r]);
      }
    }
  }
}

static inline int cdiv_h(int a, int b) { return (a + b - 1) / b; }

void conv_dw_tr(const GemmParams& p, hipStream_t s) {
  // TBM=256 halves the per-M-tile operand re-reads (gx 7 -> 4 at M=800)
  // but its 80 KB LDS drops occupancy to 2 blocks/CU and LOSES at both
  // batch sizes (114 vs 81 us @1024, 511 vs 365 @8192) — same pattern as
  // the conv dX glds experiment: these latency-bound gather kernels live
  // on TLP.  Kept instantiated for DMNIST_DW_TBM=256 A/B runs.
  static int tbm = [] {
    const char* e = getenv("DMNIST_DW_TBM");
    return e ? atoi(e) : 128;
  }();
  static int tbk = [] {
    const char* e = getenv("DMNIST_DW_TBK");
    return e ? atoi(e) : 32;  // 24 KB LDS -> 6 blocks/CU (77.8 vs 81.3 us)
  }();
  dim3 grid(cdiv_h(p.M, tbm >= 256 ? 256 : 128) * cdiv_h(p.N, 64) * p.splitk);
  if (tbm >= 256)
    hipLaunchKernelGGL((dw_tr_kernel<64, AM_CONV5, 256>), grid, dim3(NT), 0,
                       s, p);
  else if (tbk <= 32 && getenv("DMNIST_DW_SPAN"))
    // 3-buffer glds span (36 KB LDS -> 4 blocks/CU)
    hipLaunchKernelGGL((dw_tr_kernel<64, AM_CONV5, 128, 32, 3>), grid,
                       dim3(NT), 0, s, p);
  else if (tbk <= 32)
    // 24 KB LDS -> 6 blocks/CU: TLP has been the winning lever on every
    // latency-bound gather kernel in this file's history
    hipLaunchKernelGGL((dw_tr_kernel<64, AM_CONV5, 128, 32>), grid, dim3(NT),
                       0, s, p);
  else
    hipLaunchKernelGGL((dw_tr_kernel<64, AM_CONV5, 128>), grid, dim3(NT), 0,
                       s, p);
}

void gemm_dw_tr_128(const GemmParams& p, hipStream_t s) {
  dim3 grid(cdiv_h(p.M, 128) * cdiv_h(p.N, 128) * p.splitk);
  hipLaunchKernelGGL((dw_tr_kernel<128, AM_PLAIN>), grid, dim3(NT), 0, s, p);
}

void gemm_dw_tr_64(const GemmParams& p, hipStream_t s) {
  dim3 grid(cdiv_h(p.M, 128) * cdiv_h(p.N, 64) * p.splitk);
  hipLaunchKernelGGL((dw_tr_kernel<64, AM_PLAIN>), grid, dim3(NT), 0, s, p);
}
