// Implicit-GEMM MFMA tile kernel for gfx950 (CDNA4) — the single compute
// template behind every matmul-shaped op in the model:
//   fc fwd (bias/relu/dropout epilogue), fc bwd dX / dW,
//   conv5x5-SAME fwd fused with bias+ReLU+maxpool2x2 (pool-grouped M
//   ordering: m = pooled_pixel*4 + window_pos, so pooling is a per-lane
//   4-register max in the MFMA epilogue — zero cross-lane traffic),
//   conv bwd dX and dW (im2col gathers in the LDS staging stage).
//
// Structure: 256 threads = 4 waves (2x2), v_mfma_f32_16x16x32_bf16,
// BK=64, double-buffered LDS, ONE barrier per K-step with the T14
// issue-early/write-late split (cdna_hip_programming.md §6 G15): the next
// tile's global loads are issued BEFORE the current tile's MFMAs so HBM
// latency hides under compute; the ds_write lands after the barrier.
// LDS images are [outer][K+8] (the +8 bf16 row pad makes the 16-lane
// ds_read_b128 fragment reads conflict-free: row stride 144B).

#include "common.h"
#include "kernels.h"

// ---- modes ---------------------------------------------------------------
enum AMode {
  A_N = 0,         // A[m][k] = Asrc[m*lda + k]
  A_T = 1,         // A[m][k] = Asrc[k*lda + m]           (fc dW: x^T)
  A_CONV_FWD = 2,  // im2col gather, pool-grouped m, Cin % 8 == 0
  A_CONV1_FWD = 3, // im2col gather, Cin == 1 (K = 25, one K-step)
  A_CONV_DX = 4,   // dact gather: A[m=in pixel][k=(khkw,co)]
  A_CONV_DW = 5,   // x gather transposed: A[m=(khkw,ci)][k=pixel]
};
enum BMode {
  B_KMAJ = 0,      // Bmat[k][n] = Bsrc[k*ldb + n] (transpose-staged)
  B_NMAJ = 1,      // Bmat[k][n] = Bsrc[n*ldb + k] (row-staged; fc dX: W rows)
  B_CONV_DX_W = 2, // Bmat[k=(khkw,co)][n=ci] = w[((khkw)*Cin+ci)*Cout + co]
};
enum Epi { EPI_NONE = 0, EPI_BIAS = 1, EPI_BIAS_RELU = 2, EPI_BIAS_RELU_DROP = 3,
           EPI_POOL = 4, EPI_UNPOOL = 5, EPI_MASK_DB = 6 };
// EPI_MASK_DB: dX epilogue that recovers the downstream relu+dropout mask
// from the sign of the saved activation (actm) — v = a>0 ? v/p_keep : 0 —
// and accumulates the bias grad db[n] (column sums).  Folds the standalone
// relu_drop_bwd (mask_db) pass into the producing dX GEMM.
// EPI_UNPOOL: the output row m is a batch index and column n a pooled
// feature (ho, wo, c) of a [CHo][CWo][Cout] activation; instead of writing
// C[m][n], the epilogue routes the value through the maxpool-2x2 backward
// (relu-mask by ypool > 0, scatter to the argmax position, 3 zeros) straight
// into the pre-pool gradient dact [CB][2*CHo][2*CWo][Cout], and accumulates
// the conv bias grad db[c] — this FUSES the standalone pool_bwd_scatter
// kernel into the producing dX GEMM (reference backward chain
// mnist.py:115-127), removing one critical-path kernel + the pooled-grad
// round trip.
enum OutKind { OUT_BF16 = 0, OUT_F32_ATOMIC = 1, OUT_F32_SLICES = 2 };

#define NTHREADS 256
#define BK 64
#define LDK (BK + 8)

// load up to 8 bf16 from p[0..vcnt), zero-fill the rest; vector fast path
DEV short8 loadRow8(const ushort_t* p, int vcnt) {
  short8 v;
  if (vcnt >= 8 && ((uintptr_t)p & 15) == 0) {
    v = *reinterpret_cast<const short8*>(p);
  } else {
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = (j < vcnt) ? (short)p[j] : (short)0;
  }
  return v;
}

// decode pool-grouped output index m -> conv-output pixel (n, oy, ox)
DEV bool decode_pool_m(const GemmParams& p, int m, int& n, int& oy, int& ox) {
  int q = m >> 2, pos = m & 3;
  int wo = q % p.CWo;
  int t = q / p.CWo;
  int ho = t % p.CHo;
  n = t / p.CHo;
  oy = ho * 2 + (pos >> 1);
  ox = wo * 2 + (pos & 1);
  return n < p.CB;
}

// ---- staging: load phase (global -> regs), mode-specific addressing ------
// Chunk c covers either 8 k-elements of one row (rows16 layout) or 8 outer
// elements of one k (trans layout).  CH chunks per thread.

template <int BM, int AMODE, int CH>
DEV void loadA(const GemmParams& p, int m0, int kt, int kend, int tid,
               short8 (&regs)[CH]) {
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    int c = tid + j * NTHREADS;
    short8 v = short8{0, 0, 0, 0, 0, 0, 0, 0};
    if (AMODE == A_N || AMODE == A_CONV_FWD || AMODE == A_CONV_DX) {
      int i = c / (BK / 8);
      int kc = (c % (BK / 8)) * 8;
      int m = m0 + i, k = kt + kc;
      int vcnt = min(8, kend - k);
      if (m < p.M && vcnt > 0) {
        if (AMODE == A_N) {
          v = loadRow8(p.A + (size_t)m * p.lda + k, vcnt);
        } else if (AMODE == A_CONV_FWD) {
          int khkw = k / p.Cin, ci = k % p.Cin;
          int kh = khkw / 5, kw = khkw % 5;
          int n, oy, ox;
          if (decode_pool_m(p, m, n, oy, ox)) {
            int y = oy + kh - 2, x = ox + kw - 2;
            if (y >= 0 && y < p.CH && x >= 0 && x < p.CW)
              v = loadRow8(p.A + (((size_t)n * p.CH + y) * p.CW + x) * p.Cin + ci,
                           min(vcnt, p.Cin - ci));
          }
        } else {  // A_CONV_DX
          int wi = m % p.CW, t2 = m / p.CW;
          int hi = t2 % p.CH, n = t2 / p.CH;
          int khkw = k / p.Cout, co = k % p.Cout;
          int kh = khkw / 5, kw = khkw % 5;
          int y = hi - kh + 2, x = wi - kw + 2;
          if (n < p.CB && y >= 0 && y < p.CH && x >= 0 && x < p.CW)
            v = loadRow8(p.A + (((size_t)n * p.CH + y) * p.CW + x) * p.Cout + co,
                         min(vcnt, p.Cout - co));
        }
      }
    } else if (AMODE == A_T) {
      int kk = c / (BM / 8);
      int i0 = (c % (BM / 8)) * 8;
      int k = kt + kk, m = m0 + i0;
      if (k < kend) v = loadRow8(p.A + (size_t)k * p.lda + m, min(8, p.M - m));
    } else if (AMODE == A_CONV1_FWD) {
      // Cin==1, K==25: 8 scalar gathers per chunk (tiny op)
      int i = c / (BK / 8);
      int kc = (c % (BK / 8)) * 8;
      int m = m0 + i;
      int n, oy, ox;
      if (m < p.M && kc < 25 && decode_pool_m(p, m, n, oy, ox)) {
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          int k = kt + kc + e;
          if (k < kend && k < 25) {
            int kh = k / 5, kw = k % 5;
            int y = oy + kh - 2, x = ox + kw - 2;
            if (y >= 0 && y < p.CH && x >= 0 && x < p.CW)
              v[e] = p.A[((size_t)n * p.CH + y) * p.CW + x];
          }
        }
      }
    } else {  // A_CONV_DW: A[m=(khkw,ci)][k=pixel]
      int kk = c / (BM / 8);
      int i0 = (c % (BM / 8)) * 8;
      int k = kt + kk, m = m0 + i0;
      if (k < kend && m < p.M) {
        int w_ = k % p.CW, t2 = k / p.CW;
        int h_ = t2 % p.CH, n = t2 / p.CH;
        if (n < p.CB) {
          if (p.Cin % 8 == 0) {
            // 8 consecutive m share one khkw (ci0 aligned to 8): vector load
            int khkw = m / p.Cin, ci = m % p.Cin;
            int kh = khkw / 5, kw = khkw % 5;
            int y = h_ + kh - 2, x = w_ + kw - 2;
            if (y >= 0 && y < p.CH && x >= 0 && x < p.CW)
              v = loadRow8(p.A + (((size_t)n * p.CH + y) * p.CW + x) * p.Cin + ci,
                           min(8, p.M - m));
          } else {
            // Cin==1: every m is a different filter tap — per-element gather
#pragma unroll
            for (int e = 0; e < 8; ++e) {
              int me = m + e;
              if (me >= p.M) break;
              int khkw = me / p.Cin, ci = me % p.Cin;
              int kh = khkw / 5, kw = khkw % 5;
              int y = h_ + kh - 2, x = w_ + kw - 2;
              if (y >= 0 && y < p.CH && x >= 0 && x < p.CW)
                v[e] = p.A[(((size_t)n * p.CH + y) * p.CW + x) * p.Cin + ci];
            }
          }
        }
      }
    }
    regs[j] = v;
  }
}

template <int BM, int AMODE, int CH>
DEV void writeA(ushort_t (*As)[LDK], int tid, const short8 (&regs)[CH]) {
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    int c = tid + j * NTHREADS;
    if (AMODE == A_N || AMODE == A_CONV_FWD || AMODE == A_CONV_DX ||
        AMODE == A_CONV1_FWD) {
      int i = c / (BK / 8);
      int kc = (c % (BK / 8)) * 8;
      *reinterpret_cast<short8*>(&As[i][kc]) = regs[j];
    } else {  // trans layouts scatter 8 outer rows at one k column
      int kk = c / (BM / 8);
      int i0 = (c % (BM / 8)) * 8;
#pragma unroll
      for (int e = 0; e < 8; ++e) As[i0 + e][kk] = regs[j][e];
    }
  }
}

template <int BN, int BMODE, int CH>
DEV void loadB(const GemmParams& p, int n0, int kt, int kend, int tid,
               short8 (&regs)[CH]) {
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    int c = tid + j * NTHREADS;
    short8 v = short8{0, 0, 0, 0, 0, 0, 0, 0};
    if (BMODE == B_KMAJ) {
      int kk = c / (BN / 8);
      int j0 = (c % (BN / 8)) * 8;
      int k = kt + kk, n = n0 + j0;
      if (k < kend) v = loadRow8(p.B + (size_t)k * p.ldb + n, min(8, p.N - n));
    } else if (BMODE == B_NMAJ) {
      int i = c / (BK / 8);
      int kc = (c % (BK / 8)) * 8;
      int n = n0 + i, k = kt + kc;
      int vcnt = min(8, kend - k);
      if (n < p.N && vcnt > 0) v = loadRow8(p.B + (size_t)n * p.ldb + k, vcnt);
    } else {  // B_CONV_DX_W
      int i = c / (BK / 8);
      int kc = (c % (BK / 8)) * 8;
      int n = n0 + i, k = kt + kc;
      if (n < p.N && k < kend) {
        int khkw = k / p.Cout, co = k % p.Cout;
        v = loadRow8(p.B + ((size_t)khkw * p.Cin + n) * p.Cout + co,
                     min(min(8, kend - k), p.Cout - co));
      }
    }
    regs[j] = v;
  }
}

template <int BN, int BMODE, int CH>
DEV void writeB(ushort_t (*Bs)[LDK], int tid, const short8 (&regs)[CH]) {
#pragma unroll
  for (int j = 0; j < CH; ++j) {
    int c = tid + j * NTHREADS;
    if (BMODE == B_KMAJ) {
      int kk = c / (BN / 8);
      int j0 = (c % (BN / 8)) * 8;
#pragma unroll
      for (int e = 0; e < 8; ++e) Bs[j0 + e][kk] = regs[j][e];
    } else {
      int i = c / (BK / 8);
      int kc = (c % (BK / 8)) * 8;
      *reinterpret_cast<short8*>(&Bs[i][kc]) = regs[j];
    }
  }
}

// ---------------------------------------------------------------------------
// PIPE=1: double-buffered LDS, issue-early/write-late, 1 barrier/K-step —
//   wins for small grids (fc GEMMs) where occupancy is grid-limited anyway.
// PIPE=0: single buffer, 2 barriers/K-step, half the LDS — wins for the
//   huge-grid conv gathers where 4-5 blocks/CU of TLP hide latency better
//   than the in-wave pipeline (measured: conv2 fwd 103us PIPE0 vs 120us
//   PIPE1; fc1 fwd 94us PIPE0 vs 59us PIPE1).
// PIPE=2: single buffer + register-carried next tile (T14 write-after-
//   barrier): next tile's gather loads issue BEFORE the current tile's
//   MFMAs so their latency hides under compute, at PIPE=0's LDS footprint
//   (the dW GEMMs were memory-level-parallelism-bound: ~6 16B loads in
//   flight per thread serial-blocked at the ds_write vmcnt = ~2 TB/s).
template <int BM, int BN, int AMODE, int BMODE, int EPI, int OUT, int PIPE,
          int SWZ = 0>
__global__ __launch_bounds__(NTHREADS)
void gemm_tile_kernel(GemmParams p) {
  constexpr int WM = BM / 2, WN = BN / 2;
  constexpr int MI = WM / 16, NI = WN / 16;
  constexpr int CHA = (BM * BK / 8) / NTHREADS;
  constexpr int CHB = (BN * BK / 8) / NTHREADS;
  constexpr int DB = (PIPE == 1) ? 2 : 1;
  static_assert(CHA >= 1 && CHB >= 1, "tile too small for 256 threads");
  __shared__ __align__(16) ushort_t As[DB][BM][LDK];
  __shared__ __align__(16) ushort_t Bs[DB][BN][LDK];
  // EPI_UNPOOL / EPI_MASK_DB: per-column bias-grad partials, flushed once
  constexpr bool HAS_DB = (EPI == EPI_UNPOOL || EPI == EPI_MASK_DB);
  __shared__ float dbred[HAS_DB ? BN : 1];
  if (HAS_DB && threadIdx.x < BN) dbred[threadIdx.x] = 0.f;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 1, wc = wave & 1;
  int bx = blockIdx.x, by = blockIdx.y, bz = blockIdx.z;
  if (SWZ) {
    // bijective XCD-chunk remap (cdna_hip_programming.md T1): blocks with
    // the same split-K slice / N-tile land on ONE XCD so the operand slice
    // they share stays in that XCD's L2 instead of being re-fetched via L3.
    // SWZ launches use a 1-D grid (blockIdx.x only) so the observed
    // b -> XCD b%8 mapping is unambiguous; the logical (x,y,z) shape is
    // recomputed from the problem dims.
    int gx = (p.M + BM - 1) / BM;
    int gy = (p.N + BN - 1) / BN;
    int total = gx * gy * p.splitk;
    int flat = blockIdx.x;
    int q = total / 8, r = total % 8;
    int xcd = flat % 8, pos = flat / 8;
    int flat2 = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + pos;
    bx = flat2 % gx;
    int rest = flat2 / gx;
    by = rest % gy;
    bz = rest / gy;
  }
  const int m0 = bx * BM;
  const int n0 = by * BN;

  int ksteps_total = (p.K + BK - 1) / BK;
  int steps_per = (ksteps_total + p.splitk - 1) / p.splitk;
  int kbeg = bz * steps_per * BK;
  int kend = min(p.K, kbeg + steps_per * BK);
  int nt = (kend - kbeg + BK - 1) / BK;
  if (nt <= 0) nt = 0;

  f32x4 acc[MI][NI];
#pragma unroll
  for (int mi = 0; mi < MI; ++mi)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) acc[mi][ni] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int lrow = lane & 15;
  const int kq = (lane >> 4) * 8;

  short8 ra[CHA], rb[CHB];
  int cur = 0;
  if (PIPE == 2) {
    if (nt > 0) {
      loadA<BM, AMODE, CHA>(p, m0, kbeg, kend, tid, ra);
      loadB<BN, BMODE, CHB>(p, n0, kbeg, kend, tid, rb);
      writeA<BM, AMODE, CHA>(As[0], tid, ra);
      writeB<BN, BMODE, CHB>(Bs[0], tid, rb);
      __syncthreads();
    }
    for (int t = 0; t < nt; ++t) {
      const bool have_next = (t + 1) < nt;
      if (have_next) {
        int kn = kbeg + (t + 1) * BK;
        loadA<BM, AMODE, CHA>(p, m0, kn, kend, tid, ra);  // hides under MFMA
        loadB<BN, BMODE, CHB>(p, n0, kn, kend, tid, rb);
      }
#pragma unroll
      for (int kh = 0; kh < 2; ++kh) {
        short8 af[MI], bf[NI];
#pragma unroll
        for (int mi = 0; mi < MI; ++mi)
          af[mi] = *reinterpret_cast<const short8*>(
              &As[0][wr * WM + mi * 16 + lrow][kh * 32 + kq]);
#pragma unroll
        for (int ni = 0; ni < NI; ++ni)
          bf[ni] = *reinterpret_cast<const short8*>(
              &Bs[0][wc * WN + ni * 16 + lrow][kh * 32 + kq]);
#pragma unroll
        for (int mi = 0; mi < MI; ++mi)
#pragma unroll
          for (int ni = 0; ni < NI; ++ni)
            acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
      }
      if (have_next) {
        __syncthreads();   // all waves done reading the buffer
        writeA<BM, AMODE, CHA>(As[0], tid, ra);
        writeB<BN, BMODE, CHB>(Bs[0], tid, rb);
        __syncthreads();
      }
    }
    goto epilogue;
  }
  if (PIPE && nt > 0) {
    loadA<BM, AMODE, CHA>(p, m0, kbeg, kend, tid, ra);
    loadB<BN, BMODE, CHB>(p, n0, kbeg, kend, tid, rb);
    writeA<BM, AMODE, CHA>(As[0], tid, ra);
    writeB<BN, BMODE, CHB>(Bs[0], tid, rb);
    __syncthreads();
  }

  for (int t = 0; t < nt; ++t) {
    const bool have_next = PIPE && (t + 1) < nt;
    if (PIPE) {
      if (have_next) {
        int kn = kbeg + (t + 1) * BK;
        loadA<BM, AMODE, CHA>(p, m0, kn, kend, tid, ra);  // issue early (T14)
        loadB<BN, BMODE, CHB>(p, n0, kn, kend, tid, rb);
      }
    } else {
      int kt = kbeg + t * BK;
      loadA<BM, AMODE, CHA>(p, m0, kt, kend, tid, ra);
      loadB<BN, BMODE, CHB>(p, n0, kt, kend, tid, rb);
      writeA<BM, AMODE, CHA>(As[0], tid, ra);
      writeB<BN, BMODE, CHB>(Bs[0], tid, rb);
      __syncthreads();
    }
    // compute current tile: BK=64 = two K=32 MFMA sub-steps
#pragma unroll
    for (int kh = 0; kh < 2; ++kh) {
      short8 af[MI], bf[NI];
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
        af[mi] = *reinterpret_cast<const short8*>(
            &As[cur][wr * WM + mi * 16 + lrow][kh * 32 + kq]);
#pragma unroll
      for (int ni = 0; ni < NI; ++ni)
        bf[ni] = *reinterpret_cast<const short8*>(
            &Bs[cur][wc * WN + ni * 16 + lrow][kh * 32 + kq]);
#pragma unroll
      for (int mi = 0; mi < MI; ++mi)
#pragma unroll
        for (int ni = 0; ni < NI; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
    }
    if (PIPE) {
      if (have_next) {
        writeA<BM, AMODE, CHA>(As[cur ^ 1], tid, ra);  // write after MFMAs
        writeB<BN, BMODE, CHB>(Bs[cur ^ 1], tid, rb);
        __syncthreads();
        cur ^= 1;
      }
    } else {
      __syncthreads();
    }
  }

  // ---- epilogue ----------------------------------------------------------
epilogue:
  const int frow = (lane >> 4) * 4;  // C/D: row=(lane>>4)*4+reg, col=lane&15
  const int fcol = lane & 15;
  float dbloc[(EPI == EPI_UNPOOL || EPI == EPI_MASK_DB) ? NI : 1];
  if (EPI == EPI_UNPOOL || EPI == EPI_MASK_DB)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) dbloc[ni] = 0.f;
#pragma unroll
  for (int mi = 0; mi < MI; ++mi) {
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int gc = n0 + wc * WN + ni * 16 + fcol;
      if (gc >= p.N) continue;
      if (EPI == EPI_UNPOOL) {
        // n = (ho*CWo + wo)*Cout + c in exactly the pooled tensor layout,
        // so ypool/amax share the [m][n] offset; dact is the 2x up-scaled
        // grid
        int c = gc % p.Cout;
        int pix = gc / p.Cout;
        int wo = pix % p.CWo, ho = pix / p.CWo;
        ushort_t* dact = reinterpret_cast<ushort_t*>(p.C);
        const int H2 = p.CHo * 2, W2 = p.CWo * 2;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int gr = m0 + wr * WM + mi * 16 + frow + r;
          if (gr >= p.M) continue;
          size_t qoff = (size_t)gr * p.N + gc;
          int pos = p.amax[qoff];  // 0..3 live, 7 dead (no y read needed)
          float g = pos < 4 ? acc[mi][ni][r] : 0.f;
          dbloc[ni] += g;
          ushort_t gb = f2bf(g);
#pragma unroll
          for (int rr = 0; rr < 2; ++rr)
#pragma unroll
            for (int cc2 = 0; cc2 < 2; ++cc2) {
              size_t o = (((size_t)gr * H2 + ho * 2 + rr) * W2 + wo * 2 +
                          cc2) * p.Cout + c;
              dact[o] = (pos == rr * 2 + cc2) ? gb : (ushort_t)0;
            }
        }
        continue;
      }
      float bias_v = (EPI != EPI_NONE && p.bias) ? p.bias[gc] : 0.f;
      if (EPI == EPI_POOL) {
        // rows (q*4 .. q*4+3) of this lane group are one pool window
        int gq = (m0 + wr * WM + mi * 16 + frow) >> 2;
        if (gq * 4 >= p.M) continue;
        float best = -1.0f / 0.0f;
        int barg = 0;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[mi][ni][r] + bias_v;
          v = v > 0.f ? v : 0.f;  // conv bias + relu BEFORE pool
          if (v > best) { best = v; barg = r; }
        }
        ushort_t* Cb = reinterpret_cast<ushort_t*>(p.C);
        Cb[(size_t)gq * p.ldc + gc] = f2bf(best);
        if (p.amax)  // 7 = dead window (liveness for pooled-consumer bwd)
          p.amax[(size_t)gq * p.ldc + gc] = (uint8_t)(best > 0.f ? barg : 7);
      } else {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int gr = m0 + wr * WM + mi * 16 + frow + r;
          if (gr >= p.M) continue;
          float v = acc[mi][ni][r];
          if (EPI == EPI_MASK_DB) {
            float a = bf2f(p.actm[(size_t)gr * p.ldc + gc]);
            v = (a > 0.f) ? v * p.p_keep : 0.f;  // p_keep holds 1/keep here
            dbloc[ni] += v;
          }
          if (EPI != EPI_NONE && EPI != EPI_MASK_DB) v += bias_v;
          if (EPI == EPI_BIAS_RELU || EPI == EPI_BIAS_RELU_DROP)
            v = v > 0.f ? v : 0.f;
          if (EPI == EPI_BIAS_RELU_DROP) {
            uint64_t off = p.offset_dev ? (uint64_t)*p.offset_dev : p.offset;
            float u = philox_uniform(p.seed, off, (uint64_t)gr * p.N + gc);
            v = (u < p.p_keep) ? v / p.p_keep : 0.f;
          }
          if (OUT == OUT_BF16) {
            reinterpret_cast<ushort_t*>(p.C)[(size_t)gr * p.ldc + gc] = f2bf(v);
          } else if (OUT == OUT_F32_SLICES) {
            // split-K without atomics: each k-slice owns a full [M][N]
            // plane; a deterministic epilogue sums the planes (fwd path
            // must be bitwise-reproducible per (seed, step))
            float* outp = reinterpret_cast<float*>(p.C) +
                          (size_t)bz * p.M * p.ldc;
            outp[(size_t)gr * p.ldc + gc] = v;
          } else {
            atomicAdd(reinterpret_cast<float*>(p.C) + (size_t)gr * p.ldc + gc, v);
          }
        }
      }
    }
  }
  if (EPI == EPI_UNPOOL || EPI == EPI_MASK_DB) {
    // bias-grad flush: lane partials -> LDS per-column -> one global
    // atomicAdd per block column (UNPOOL: channel = col % Cout)
#pragma unroll
    for (int ni = 0; ni < NI; ++ni) {
      int colb = wc * WN + ni * 16 + fcol;
      if (dbloc[ni] != 0.f) atomicAdd(&dbred[colb], dbloc[ni]);
    }
    __syncthreads();
    if (p.db && (int)threadIdx.x < BN) {
      int gc = n0 + (int)threadIdx.x;
      float v = dbred[threadIdx.x];
      int idx = (EPI == EPI_UNPOOL) ? gc % p.Cout : gc;
      if (gc < p.N && v != 0.f) atomicAdd(&p.db[idx], v);
    }
  }
}

// ---- host-side launch table ----------------------------------------------
#include <hip/hip_runtime_api.h>

static inline int cdiv_host(int a, int b) { return (a + b - 1) / b; }

// Explicit entry points used by bindings.cpp (keeps instantiations bounded).
#define GEMM_ENTRY(name, BM, BN, AM, BMo, EPI, OUT, PIPE)                   \
  void name(const GemmParams& p, hipStream_t s) {                            \
    dim3 grid(cdiv_host(p.M, BM), cdiv_host(p.N, BN), p.splitk);             \
    hipLaunchKernelGGL((gemm_tile_kernel<BM, BN, AM, BMo, EPI, OUT, PIPE>),  \
                       grid, dim3(NTHREADS), 0, s, p);                       \
  }
#define GEMM_ENTRY_SWZ(name, BM, BN, AM, BMo, EPI, OUT, PIPE)                \
  void name(const GemmParams& p, hipStream_t s) {                            \
    dim3 grid(cdiv_host(p.M, BM) * cdiv_host(p.N, BN) * p.splitk);           \
    hipLaunchKernelGGL((gemm_tile_kernel<BM, BN, AM, BMo, EPI, OUT, PIPE,    \
                                         1>),                                \
                       grid, dim3(NTHREADS), 0, s, p);                       \
  }

// fc GEMMs: grids of 32-400 WGs (occupancy grid-limited) -> PIPE=1
GEMM_ENTRY(gemm_fwd_bias_128, 128, 128, A_N, B_KMAJ, EPI_BIAS, OUT_BF16, 1)
GEMM_ENTRY(gemm_fwd_bias_64, 64, 64, A_N, B_KMAJ, EPI_BIAS, OUT_BF16, 1)
GEMM_ENTRY(gemm_fwd_relu_128, 128, 128, A_N, B_KMAJ, EPI_BIAS_RELU, OUT_BF16, 1)
GEMM_ENTRY(gemm_fwd_relu_64, 64, 64, A_N, B_KMAJ, EPI_BIAS_RELU, OUT_BF16, 1)
GEMM_ENTRY(gemm_fwd_drop_128, 128, 128, A_N, B_KMAJ, EPI_BIAS_RELU_DROP, OUT_BF16, 1)
GEMM_ENTRY(gemm_fwd_drop_64, 64, 64, A_N, B_KMAJ, EPI_BIAS_RELU_DROP, OUT_BF16, 1)
GEMM_ENTRY(gemm_dx_128, 128, 128, A_N, B_NMAJ, EPI_NONE, OUT_BF16, 1)
GEMM_ENTRY(gemm_dx_64, 64, 64, A_N, B_NMAJ, EPI_NONE, OUT_BF16, 1)
// dX fused with maxpool-2x2 backward (fc1 dX -> dact2 directly + conv2 db)
GEMM_ENTRY(gemm_dx_unpool_128, 128, 128, A_N, B_NMAJ, EPI_UNPOOL, OUT_BF16, 1)
GEMM_ENTRY(gemm_dx_unpool_64, 64, 64, A_N, B_NMAJ, EPI_UNPOOL, OUT_BF16, 1)
// dX fused with the downstream relu/dropout mask + bias-grad column sums
GEMM_ENTRY(gemm_dx_mask_128, 128, 128, A_N, B_NMAJ, EPI_MASK_DB, OUT_BF16, 1)
GEMM_ENTRY(gemm_dx_mask_64, 64, 64, A_N, B_NMAJ, EPI_MASK_DB, OUT_BF16, 1)
GEMM_ENTRY_SWZ(gemm_dw_128, 128, 128, A_T, B_KMAJ, EPI_NONE, OUT_F32_ATOMIC, 2)
GEMM_ENTRY_SWZ(gemm_dw_64, 64, 64, A_T, B_KMAJ, EPI_NONE, OUT_F32_ATOMIC, 2)
// B-transposed (pre-transposed weight) fwd variants: ldb = K, vector staging
GEMM_ENTRY(gemm_fwd_bias_128_bt, 128, 128, A_N, B_NMAJ, EPI_BIAS, OUT_BF16, 1)
GEMM_ENTRY(gemm_fwd_bias_64_bt, 64, 64, A_N, B_NMAJ, EPI_BIAS, OUT_BF16, 1)
GEMM_ENTRY(gemm_fwd_relu_128_bt, 128, 128, A_N, B_NMAJ, EPI_BIAS_RELU, OUT_BF16, 1)
GEMM_ENTRY(gemm_fwd_relu_64_bt, 64, 64, A_N, B_NMAJ, EPI_BIAS_RELU, OUT_BF16, 1)
GEMM_ENTRY(gemm_fwd_drop_128_bt, 128, 128, A_N, B_NMAJ, EPI_BIAS_RELU_DROP, OUT_BF16, 1)
GEMM_ENTRY(gemm_fwd_drop_64_bt, 64, 64, A_N, B_NMAJ, EPI_BIAS_RELU_DROP, OUT_BF16, 1)
// split-K fwd slices (grid-starved small-M cases): deterministic planes
GEMM_ENTRY(gemm_fwd_slices_64, 64, 64, A_N, B_KMAJ, EPI_NONE, OUT_F32_SLICES, 1)
GEMM_ENTRY(gemm_fwd_slices_64_bt, 64, 64, A_N, B_NMAJ, EPI_NONE, OUT_F32_SLICES, 1)
// conv gathers: thousands of WGs -> single-buffer, TLP hides latency
GEMM_ENTRY(conv_fwd_pool, 128, 64, A_CONV_FWD, B_KMAJ, EPI_POOL, OUT_BF16, 0)
GEMM_ENTRY(conv_fwd_pool_bt, 128, 64, A_CONV_FWD, B_NMAJ, EPI_POOL, OUT_BF16, 0)
GEMM_ENTRY(conv1_fwd_pool, 128, 64, A_CONV1_FWD, B_KMAJ, EPI_POOL, OUT_BF16, 0)
GEMM_ENTRY(conv_dx_gemm, 128, 32, A_CONV_DX, B_CONV_DX_W, EPI_NONE, OUT_BF16, 0)
// conv dW: gather-staged -> occupancy (PIPE=0) beats the in-wave pipeline
// (measured 2069us PIPE=1 vs 1309 at BM=64 PIPE=0); BM=128 halves the
// per-M-tile dact re-reads
GEMM_ENTRY_SWZ(conv_dw_gemm, 128, 64, A_CONV_DW, B_KMAJ, EPI_NONE, OUT_F32_ATOMIC, 2)
// conv1 dW: M=25 -> 32x32 tile (78% M-utilization vs 39% at BM=64)
GEMM_ENTRY_SWZ(conv1_dw_gemm, 32, 32, A_CONV_DW, B_KMAJ, EPI_NONE, OUT_F32_ATOMIC, 2)
