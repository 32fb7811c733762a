// Non-GEMM fused kernels: relu/dropout gradient mask (+ bias-grad column
// sums), maxpool backward scatter (+ conv bias grad), fused softmax-CE
// (+top-1 correct count), the fused flat SGD apply (+drop-connect mask,
// + bf16 shadow refresh), bf16 transpose, and the Cin=1 conv layer's
// direct v_dot2c_f32_bf16 forward/dW (K=25 is below MFMA's win threshold).

#include "common.h"
#include "kernels.h"

// ---------------------------------------------------------------------------
// dyeff[b,n] = dy * (y>0) / p_keep ; db[n] += column sums (fp32)
// Recovers BOTH the relu and the dropout mask from sign(y)
// (ops/functional.py LinearActFn docstring).
// Block: 256 threads as 4 rows x 64 cols; grid (col_groups, row_slices).
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256)
void relu_drop_bwd_kernel(const ushort_t* dy, const ushort_t* y,
                          ushort_t* dyeff, float* db, int B, int N,
                          float inv_keep, int apply_mask, int rows_per_slice) {
  __shared__ float partial[64];
  int cg = blockIdx.x;           // 64-column group
  int rs = blockIdx.y;           // row slice
  int c = (threadIdx.x & 63);
  int rlane = threadIdx.x >> 6;  // 0..3
  int n = cg * 64 + c;
  float sum = 0.f;
  if (n < N) {
    int r0 = rs * rows_per_slice;
    int r1 = min(B, r0 + rows_per_slice);
    for (int r = r0 + rlane; r < r1; r += 4) {
      size_t idx = (size_t)r * N + n;
      float g = bf2f(dy[idx]);
      if (apply_mask) g = (bf2f(y[idx]) > 0.f) ? g * inv_keep : 0.f;
      dyeff[idx] = f2bf(g);
      sum += g;
    }
  }
  // reduce 4 row-lanes per column through LDS
  if (rlane == 0) partial[c] = 0.f;
  __syncthreads();
  atomicAdd(&partial[c], sum);
  __syncthreads();
  if (rlane == 0 && n < N && db) atomicAdd(&db[n], partial[c]);
}

// ---------------------------------------------------------------------------
// Maxpool 2x2 backward scatter: route dy (masked by relu: pooled y > 0)
// to the argmax position of each window; other 3 positions zero.
// Also accumulates db[c] (conv bias grad = sum over dact).
// One thread per pooled element (q, c); dact is [B, H, W, C] bf16.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256)
void pool_bwd_scatter_kernel(const ushort_t* dy, const ushort_t* y,
                             const uint8_t* amax, ushort_t* dact, float* db,
                             int Mpool, int C, int H, int W, int Wo) {
  // one thread handles 8 consecutive channels of one pooled window; all
  // reads/writes are 16B (C % 8 == 0).  The grid is sized so each thread
  // walks >=4 granules (single-shot threads were pure latency-bound:
  // 0.8-1.5 TB/s measured); index math is 32-bit with CB8 a power of two.
  __shared__ float partial[64];
  if (threadIdx.x < 64) partial[threadIdx.x] = 0.f;
  __syncthreads();
  const unsigned CB8 = (unsigned)C / 8;
  const unsigned cb8_sh = (CB8 == 8) ? 3u : (CB8 == 4 ? 2u : 0u);
  const bool cb8_pow2 = (CB8 & (CB8 - 1)) == 0;
  unsigned total = (unsigned)Mpool * CB8;
  unsigned stride = gridDim.x * blockDim.x;
  float local[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  int mycol = -1;
  const unsigned Ho = (unsigned)H / 2;
  for (unsigned i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    unsigned cb, q;
    if (cb8_pow2 && cb8_sh) { cb = i & (CB8 - 1); q = i >> cb8_sh; }
    else { cb = i % CB8; q = i / CB8; }
    int c0 = (int)cb * 8;
    unsigned wo = q % (unsigned)Wo;
    unsigned t = q / (unsigned)Wo;
    unsigned ho = t % Ho;
    unsigned n = t / Ho;
    size_t base = (size_t)q * C + c0;
    short8 dyv = *reinterpret_cast<const short8*>(dy + base);
    short8 yv = *reinterpret_cast<const short8*>(y + base);
    uint64_t am8;
    __builtin_memcpy(&am8, amax + base, 8);
    float g[8];
    int pos[8];
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      g[e] = bf2f((ushort_t)dyv[e]);
      if (!(bf2f((ushort_t)yv[e]) > 0.f)) g[e] = 0.f;
      pos[e] = (int)((am8 >> (8 * e)) & 0xff);
      local[e] += g[e];
    }
#pragma unroll
    for (int r = 0; r < 2; ++r)
#pragma unroll
      for (int cx = 0; cx < 2; ++cx) {
        int p4 = r * 2 + cx;
        short8 out;
#pragma unroll
        for (int e = 0; e < 8; ++e)
          out[e] = (pos[e] == p4) ? (short)f2bf(g[e]) : (short)0;
        size_t o = (((size_t)n * H + ho * 2 + r) * W + wo * 2 + cx) * C + c0;
        *reinterpret_cast<short8*>(dact + o) = out;
      }
    if (mycol < 0) mycol = c0;  // stride multiple of CB8 -> c0 fixed
  }
  if (mycol >= 0) {
#pragma unroll
    for (int e = 0; e < 8; ++e)
      atomicAdd(&partial[(mycol + e) & 63], local[e]);
  }
  __syncthreads();
  if (db && threadIdx.x < 64) {
    float v = partial[threadIdx.x];
    if (C >= 64) {
      if (v != 0.f) atomicAdd(&db[threadIdx.x], v);
    } else if ((int)threadIdx.x < C) {
      float v2 = partial[threadIdx.x + 32];
      if (v + v2 != 0.f) atomicAdd(&db[threadIdx.x], v + v2);
    }
  }
}

// ---------------------------------------------------------------------------
// Fused softmax cross-entropy fwd(+grad) + top-1 correct count.
// One thread per row (C <= 16); out[0] += sum(loss)/B ; out[1] += correct.
// dlogits = (softmax - onehot)/B, bf16.
// ---------------------------------------------------------------------------
template <int CC>
__global__ __launch_bounds__(256)
void softmax_xent_t_kernel(const ushort_t* logits, const long* labels,
                           ushort_t* dlogits, float* out, int B, int C,
                           float* db, float inv_n) {
  // CC > 0: compile-time class count — every per-class loop unrolls and
  // v[] lives in REGISTERS.  The runtime-C form kept v[16] in scratch
  // (runtime-indexed array): measured 30 us at B=1024 vs ~5 with CC=10.
  // db (optional): column sums of dlogits = the fc2 bias grad, reduced
  // across the wave with shfl (one LDS atomic per class per wave — the
  // naive per-thread LDS atomics serialized 64-way).
  // inv_n: out[1] += correct * inv_n (1/B = accuracy mean in-kernel).
  if (CC) C = CC;
  __shared__ float red[2];
  __shared__ float dbred[16];
  if (threadIdx.x == 0) { red[0] = 0.f; red[1] = 0.f; }
  if (db && threadIdx.x < 16) dbred[threadIdx.x] = 0.f;
  __syncthreads();
  int b = blockIdx.x * blockDim.x + threadIdx.x;
  float loss = 0.f, correct = 0.f;
  float d[CC ? CC : 16];
  const bool live = b < B;
  {
    float v[CC ? CC : 16];
    float mx = -1e30f;
    int arg = 0;
    if (CC == 10 && live) {
      // one 16B + one 4B load per row (20B, always 4B-aligned) instead of
      // 10 scalar u16 loads
      ushort_t row[10];
      __builtin_memcpy(&row[0], logits + (size_t)b * 10, 16);
      __builtin_memcpy(&row[8], logits + (size_t)b * 10 + 8, 4);
#pragma unroll
      for (int c = 0; c < 10; ++c) {
        v[c] = bf2f(row[c]);
        if (v[c] > mx) { mx = v[c]; arg = c; }
      }
    } else {
#pragma unroll
      for (int c = 0; c < (CC ? CC : 16); ++c) {
        if (!CC && c >= C) break;
        v[c] = live ? bf2f(logits[(size_t)b * C + c]) : 0.f;
        if (v[c] > mx) { mx = v[c]; arg = c; }
      }
    }
    float se = 0.f;
#pragma unroll
    for (int c = 0; c < (CC ? CC : 16); ++c) {
      if (!CC && c >= C) break;
      v[c] = __expf(v[c] - mx);
      se += v[c];
    }
    float inv_se = 1.f / se;
    long lab = live ? labels[b] : 0;
    float invB = 1.f / (float)B;
    if (CC == 10) {
      ushort_t rowo[10];
#pragma unroll
      for (int c = 0; c < 10; ++c) {
        float pp = v[c] * inv_se;
        d[c] = live ? (pp - (c == (int)lab ? 1.f : 0.f)) * invB : 0.f;
        rowo[c] = f2bf(d[c]);
      }
      if (live) {
        __builtin_memcpy(dlogits + (size_t)b * 10, &rowo[0], 16);
        __builtin_memcpy(dlogits + (size_t)b * 10 + 8, &rowo[8], 4);
      }
    } else {
#pragma unroll
      for (int c = 0; c < (CC ? CC : 16); ++c) {
        if (!CC && c >= C) break;
        float pp = v[c] * inv_se;
        d[c] = live ? (pp - (c == (int)lab ? 1.f : 0.f)) * invB : 0.f;
        if (live) dlogits[(size_t)b * C + c] = f2bf(d[c]);
      }
    }
    if (live) {
      loss = -(__logf(v[(int)lab] * inv_se)) * invB;
      correct = (arg == (int)lab) ? 1.f : 0.f;
    }
  }
  atomicAdd(&red[0], loss);
  atomicAdd(&red[1], correct * inv_n);
  if (db) {
#pragma unroll
    for (int c = 0; c < (CC ? CC : 16); ++c) {
      if (!CC && c >= C) break;
      // wave-reduce the column contribution, one LDS atomic per wave
      float dsum = d[c];
#pragma unroll
      for (int off = 32; off > 0; off >>= 1)
        dsum += __shfl_down(dsum, off, 64);
      if ((threadIdx.x & 63) == 0 && dsum != 0.f)
        atomicAdd(&dbred[c], dsum);
    }
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    atomicAdd(&out[0], red[0]);
    atomicAdd(&out[1], red[1]);
  }
  if (db && (int)threadIdx.x < C) {
    float v2 = dbred[threadIdx.x];
    if (v2 != 0.f) atomicAdd(&db[threadIdx.x], v2);
  }
}

// ---------------------------------------------------------------------------
// Fused flat SGD apply: master -= lr*scale*(g [* bernoulli(keep)]);
// shadow = bf16(master).  Drop-connect = reference distributed_train.py:414
// (mask, NO rescale).  Vectorized float4 path + scalar tail.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256)
void sgd_step_kernel(float* master, float* grad, ushort_t* shadow,
                     int has_shadow, long n, float lr_scale, float dc_keep,
                     uint64_t seed, uint64_t offset,
                     const float* lr_scale_dev, const long* offset_dev,
                     float* momentum, float mu, int zero_grad) {
  // zero_grad: clear the flat gradient bucket in the SAME pass that
  // consumes it (the captured graph then needs no per-step fill kernel;
  // all dW/db writers accumulate with atomics into a zeroed bucket)
  // momentum (optional): v = mu*v + g; w -= lr*v  (v=nullptr => plain SGD,
  // the reference's GradientDescentOptimizer semantics)
  if (lr_scale_dev) lr_scale = *lr_scale_dev;
  if (offset_dev) offset = (uint64_t)*offset_dev;
  long i = (long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i * 4 < n; i += stride) {
    long base = i * 4;
    if (base + 3 < n) {
      float4 g = *reinterpret_cast<const float4*>(grad + base);
      float4 m = *reinterpret_cast<const float4*>(master + base);
      if (dc_keep > 0.f) {
        Philox4 ph = philox4x32(seed, offset, (uint64_t)i);
        const float s = 1.0f / 4294967296.0f;
        g.x *= (ph.x * s < dc_keep) ? 1.f : 0.f;
        g.y *= (ph.y * s < dc_keep) ? 1.f : 0.f;
        g.z *= (ph.z * s < dc_keep) ? 1.f : 0.f;
        g.w *= (ph.w * s < dc_keep) ? 1.f : 0.f;
      }
      if (momentum) {
        float4 v = *reinterpret_cast<const float4*>(momentum + base);
        v.x = mu * v.x + g.x; v.y = mu * v.y + g.y;
        v.z = mu * v.z + g.z; v.w = mu * v.w + g.w;
        *reinterpret_cast<float4*>(momentum + base) = v;
        g = v;
      }
      m.x -= lr_scale * g.x; m.y -= lr_scale * g.y;
      m.z -= lr_scale * g.z; m.w -= lr_scale * g.w;
      *reinterpret_cast<float4*>(master + base) = m;
      if (zero_grad)
        *reinterpret_cast<float4*>(grad + base) = float4{0.f, 0.f, 0.f, 0.f};
      if (has_shadow) {
        shadow[base] = f2bf(m.x); shadow[base + 1] = f2bf(m.y);
        shadow[base + 2] = f2bf(m.z); shadow[base + 3] = f2bf(m.w);
      }
    } else {
      Philox4 ph = philox4x32(seed, offset, (uint64_t)i);
      const float s = 1.0f / 4294967296.0f;
      float u[4] = {ph.x * s, ph.y * s, ph.z * s, ph.w * s};
      for (long j = base; j < n; ++j) {
        float g = grad[j];
        if (dc_keep > 0.f) g *= (u[j - base] < dc_keep) ? 1.f : 0.f;
        if (momentum) {
          float v = mu * momentum[j] + g;
          momentum[j] = v;
          g = v;
        }
        float m = master[j] - lr_scale * g;
        master[j] = m;
        if (zero_grad) grad[j] = 0.f;
        if (has_shadow) shadow[j] = f2bf(m);
      }
    }
  }
}

// step/LR advance: device-side staircase LR so a captured graph needs no
// per-step host argument updates (lr = lr0*decay^(step/decay_steps),
// folded with 1/contributors; then step++)
extern "C" __global__ void step_advance_kernel(long* step_dev,
                                               float* lr_scale_dev, float lr0,
                                               float decay, int decay_steps,
                                               float inv_contrib) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    // prime step s+1: the staircase exponent must be the NEXT step's
    // (eager lr_at(step) parity at decay boundaries)
    long s = *step_dev + 1;
    float lr = lr0 * __powf(decay, (float)(s / decay_steps));
    *lr_scale_dev = lr * inv_contrib;
    *step_dev = s;
  }
}

// batched variant: all per-step weight re-transposes in ONE launch (three
// separate ~5us launches showed up as ~20us of the B=1024 graph replay)
extern "C" __global__ __launch_bounds__(256)
void transpose_bf16_batch_kernel(TransposeBatchArgs a) {
  __shared__ ushort_t tile[32][33];
  int t = blockIdx.x;
  if (a.do_advance && t == 0 && threadIdx.x == 0) {
    // fold the step/LR advance into this (last-in-tail) launch: one fewer
    // dispatch per replayed graph; nothing later in THIS replay reads
    // step_dev/lr_scale_dev, so ordering vs the transposes is free
    long sp = *a.step_dev + 1;
    float lr = a.lr0 * __powf(a.decay, (float)(sp / a.decay_steps));
    *a.lr_scale_dev = lr * a.inv_contrib;
    *a.step_dev = sp;
  }
  int which = 0;
  while (which + 1 < a.n && t >= a.tile0[which + 1]) ++which;
  int local = t - a.tile0[which];
  const TransposeDesc d = a.d[which];
  int ctiles = (d.C + 31) / 32;
  int tr0 = (local / ctiles) * 32;
  int tc0 = (local % ctiles) * 32;
  int lx = threadIdx.x & 31, ly = threadIdx.x >> 5;
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    int r = tr0 + ly + rr * 8, c = tc0 + lx;
    tile[ly + rr * 8][lx] = (r < d.R && c < d.C)
        ? d.src[(size_t)r * d.C + c] : (ushort_t)0;
  }
  __syncthreads();
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    int c = tc0 + ly + rr * 8, r = tr0 + lx;
    if (c < d.C && r < d.R) d.dst[(size_t)c * d.R + r] = tile[lx][ly + rr * 8];
  }
}

// ---- host wrappers --------------------------------------------------------

static inline int cdivh(long a, long b) { return (int)((a + b - 1) / b); }

void launch_relu_drop_bwd(const unsigned short* dy, const unsigned short* y,
                          unsigned short* dyeff, float* db, int B, int N,
                          float inv_keep, int apply_mask, hipStream_t s) {
  // enough row-slices to fill the chip (256 CUs); 64 atomics per block on db
  int rows_per_slice = 32;
  while ((long)cdivh(N, 64) * cdivh(B, rows_per_slice) > 4096 &&
         rows_per_slice < B)
    rows_per_slice *= 2;
  dim3 grid(cdivh(N, 64), cdivh(B, rows_per_slice));
  hipLaunchKernelGGL(relu_drop_bwd_kernel, grid, dim3(256), 0, s, dy, y,
                     dyeff, db, B, N, inv_keep, apply_mask, rows_per_slice);
}

extern "C" __global__ void pool_bwd_gather_kernel(
    const ushort_t* dy, const ushort_t* y, const uint8_t* amax,
    ushort_t* dact, float* db, int Mpool, int C, int H, int W, int Wo);

void launch_pool_bwd_scatter(const unsigned short* dy, const unsigned short* y,
                             const uint8_t* amax, unsigned short* dact,
                             float* db, int Mpool, int C, int H, int W, int Wo,
                             hipStream_t s) {
  // scatter form wins (measured: gather's 4x-redundant pooled reads cost
  // more than its dense stores gain — 50.8 vs 23.1 us pool2 @1024);
  // DMNIST_POOL_GATHER=1 switches for A/B
  static int use_scatter = [] {
    const char* e = getenv("DMNIST_POOL_GATHER");
    return e ? !atoi(e) : 1;
  }();
  long total = (long)Mpool * (C / 8) * (use_scatter ? 1 : 4);
  // >=4 granules per thread for memory-level parallelism; >=256 blocks to
  // fill the chip; <=2048 to bound the db atomics
  int blocks = cdivh(total, 256 * 4);
  if (blocks > 2048) blocks = 2048;
  if (blocks < 256) blocks = cdivh(total, 256) < 256 ? cdivh(total, 256) : 256;
  hipLaunchKernelGGL(use_scatter ? pool_bwd_scatter_kernel
                                 : pool_bwd_gather_kernel,
                     dim3(blocks), dim3(256), 0, s, dy, y, amax, dact, db,
                     Mpool, C, H, W, Wo);
}

void launch_softmax_xent(const unsigned short* logits, const long* labels,
                         unsigned short* dlogits, float* out, int B, int C,
                         float* db, float inv_n, hipStream_t s) {
  // one row per thread: B=1024 in 256-thread blocks was only 4 blocks
  // (256-CU chip ~idle) — 64-thread blocks spread it; at large B the
  // extra per-block out[] atomics cost more than the spread gains
  int bt = B >= 2048 ? 256 : 64;
  dim3 grid(cdivh(B, bt));
  if (C == 10)
    hipLaunchKernelGGL((softmax_xent_t_kernel<10>), grid, dim3(bt), 0, s,
                       logits, labels, dlogits, out, B, C, db, inv_n);
  else
    hipLaunchKernelGGL((softmax_xent_t_kernel<0>), grid, dim3(bt), 0, s,
                       logits, labels, dlogits, out, B, C, db, inv_n);
}

void launch_sgd_step(float* master, float* grad, unsigned short* shadow,
                     int has_shadow, long n, float lr_scale, float dc_keep,
                     uint64_t seed, uint64_t offset, float* momentum, float mu,
                     hipStream_t s) {
  long groups = (n + 3) / 4;
  int blocks = (int)min((long)2048, (groups + 255) / 256);
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(sgd_step_kernel, dim3(blocks), dim3(256), 0, s, master,
                     grad, shadow, has_shadow, n, lr_scale, dc_keep, seed,
                     offset, (const float*)nullptr, (const long*)nullptr,
                     momentum, mu, 0);
}

void launch_sgd_step_dev(float* master, float* grad,
                         unsigned short* shadow, int has_shadow, long n,
                         const float* lr_scale_dev, float dc_keep,
                         uint64_t seed, const long* offset_dev,
                         float* momentum, float mu, int zero_grad,
                         hipStream_t s) {
  long groups = (n + 3) / 4;
  int blocks = (int)min((long)2048, (groups + 255) / 256);
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(sgd_step_kernel, dim3(blocks), dim3(256), 0, s, master,
                     grad, shadow, has_shadow, n, 0.f, dc_keep, seed, 0,
                     lr_scale_dev, offset_dev, momentum, mu, zero_grad);
}

// ---------------------------------------------------------------------------
// Per-rank PRE-AGGREGATION drop-connect mask (reference
// distributed_train.py:194-203: each worker masks its OWN gradient before
// the aggregation; masks are rank-distinct).  In-place Bernoulli(keep)
// philox mask of a flat fp32 gradient (or a slice of it at element offset
// `base`, so masking the fc/conv bucket slices separately reproduces
// exactly the whole-buffer mask).  Stream = (seed ^ SALT, step*2^20+rank):
// disjoint from both the dropout stream (offset=step) and the
// post-aggregation drop-connect stream in sgd_step.  `step_dev` (optional)
// makes it hipGraph-capturable.  Mask, NO rescale, per reference :414-416.
// ---------------------------------------------------------------------------
#define DMNIST_DC_SALT 0x9D5AD0C5u

extern "C" __global__ __launch_bounds__(256)
void grad_mask_kernel(float* g, long n, long base, float keep,
                      uint64_t seed, uint64_t step, uint64_t rank,
                      const long* step_dev) {
  if (step_dev) step = (uint64_t)*step_dev;
  uint64_t offset = (step << 20) + rank + 1;
  seed ^= (uint64_t)DMNIST_DC_SALT;
  long i = (long)(blockIdx.x) * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  const float s = 1.0f / 4294967296.0f;
  for (; i * 4 < n; i += stride) {
    long b = i * 4;
    // philox counter indexed by ABSOLUTE flat position so slice-wise
    // launches compose to the same mask as one whole-buffer launch;
    // requires base % 4 == 0 (checked host-side)
    Philox4 ph = philox4x32(seed, offset, (uint64_t)((base + b) >> 2));
    if (b + 3 < n) {
      float4 v = *reinterpret_cast<const float4*>(g + b);
      v.x *= (ph.x * s < keep) ? 1.f : 0.f;
      v.y *= (ph.y * s < keep) ? 1.f : 0.f;
      v.z *= (ph.z * s < keep) ? 1.f : 0.f;
      v.w *= (ph.w * s < keep) ? 1.f : 0.f;
      *reinterpret_cast<float4*>(g + b) = v;
    } else {
      float u[4] = {ph.x * s, ph.y * s, ph.z * s, ph.w * s};
      for (long j = b; j < n; ++j)
        g[j] *= (u[j - b] < keep) ? 1.f : 0.f;
    }
  }
}

void launch_grad_mask(float* g, long n, long base, float keep, uint64_t seed,
                      uint64_t step, uint64_t rank, const long* step_dev,
                      hipStream_t s) {
  long groups = (n + 3) / 4;
  int blocks = (int)min((long)2048, (groups + 255) / 256);
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(grad_mask_kernel, dim3(blocks), dim3(256), 0, s, g, n,
                     base, keep, seed, step, rank, step_dev);
}

void launch_step_advance(long* step_dev, float* lr_scale_dev, float lr0,
                         float decay, int decay_steps, float inv_contrib,
                         hipStream_t s) {
  hipLaunchKernelGGL(step_advance_kernel, dim3(1), dim3(64), 0, s, step_dev,
                     lr_scale_dev, lr0, decay, decay_steps < 1 ? 1 : decay_steps,
                     inv_contrib);
}

// ---------------------------------------------------------------------------
// bf16 2-D transpose dst[C][R] = src[R][C]^T — LDS 32x32 tiles (+1 pad),
// refreshes the transposed forward-GEMM weight copies after each SGD apply
// (a k-major B operand would otherwise need scatter ds_writes every K-step
// of every forward GEMM; reading a pre-transposed copy is vector-everything).
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256)
void transpose_bf16_kernel(const ushort_t* src, ushort_t* dst, int R, int C) {
  __shared__ ushort_t tile[32][33];
  int tr0 = blockIdx.y * 32;  // row block in src
  int tc0 = blockIdx.x * 32;  // col block in src
  // load 32x32 tile: 256 threads, 4 rows each of 8 cols... use 32x8 layout
  int lx = threadIdx.x & 31, ly = threadIdx.x >> 5;  // 32 x 8
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    int r = tr0 + ly + rr * 8, c = tc0 + lx;
    tile[ly + rr * 8][lx] = (r < R && c < C) ? src[(size_t)r * C + c] : 0;
  }
  __syncthreads();
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    int c = tc0 + ly + rr * 8, r = tr0 + lx;  // transposed coords
    if (c < C && r < R) dst[(size_t)c * R + r] = tile[lx][ly + rr * 8];
  }
}

// ---------------------------------------------------------------------------
// Direct conv1 fwd (Cin==1): fused 5x5 conv + bias + ReLU + maxpool2x2.
// K=25 is too small for MFMA to win (<=25/32 K-utilization plus a scalar
// im2col gather) — a VALU kernel with LDS-staged windows/weights is ~9x
// faster than the implicit-GEMM form at B=8192 (measured 579us -> VALU).
// Block: 256 threads = 8 pooled pixels x 32 Cout; each thread does
// 4 conv positions x 25 MACs fp32 from LDS broadcasts.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256)
void conv1_direct_fwd_kernel(const ushort_t* x, const ushort_t* w,
                             const float* bias, ushort_t* y, uint8_t* amax,
                             int NB, int H, int W, int Cout, int Mpool) {
  // One IMAGE per block (H=W=28): x staged once as a PACKED bf16 [32][32]
  // padded slab (2 KB); 256 threads = 32 co x 8 slots, each slot walking a
  // CONTIGUOUS run of pooled pixels.  K=25 with Cin=1 is too small for
  // MFMA; the math runs on v_dot2c_f32_bf16 (2 MACs/op): the 6-column
  // window is held as 3 even-aligned bf16 PAIRS per row (one ds_read_b32
  // each), odd-based pixels read the cross pairs built with one
  // v_alignbit, and the kw=4 tail is a dot2 against a zero-padded tap
  // pair — 15 dot2 per conv pixel, no scalar extraction anywhere.
  __shared__ ushort_t slab[32][32];
  const int Ho = H / 2, Wo = W / 2;
  const int tid = threadIdx.x;
  const int img = blockIdx.x;
  const int qpi = Ho * Wo;              // pooled pixels per image (196)
  const ushort_t* xi = x + (size_t)img * H * W;
  for (int i = tid; i < 32 * 32; i += 256) {
    int xx = i % 32, yy = i / 32;
    int sy = yy - 2, sx = xx - 2;
    ushort_t v = 0;
    if (sy >= 0 && sy < H && sx >= 0 && sx < W) v = xi[(size_t)sy * W + sx];
    slab[yy][xx] = v;
  }
  __syncthreads();
  const int co = tid & 31, slot = tid >> 5;
  if (co >= Cout) return;
  // packed taps: per kh, even pairs (w0,w1)(w2,w3) + tail pairs (w4,0) for
  // even-based pixels and (0,w4) for odd-based ones
  unsigned wpk[5][2], wt4e[5], wt4o[5];
#pragma unroll
  for (int kh = 0; kh < 5; ++kh) {
    unsigned t0 = w[(kh * 5 + 0) * Cout + co], t1 = w[(kh * 5 + 1) * Cout + co];
    unsigned t2 = w[(kh * 5 + 2) * Cout + co], t3 = w[(kh * 5 + 3) * Cout + co];
    unsigned t4 = w[(kh * 5 + 4) * Cout + co];
    wpk[kh][0] = t0 | (t1 << 16);
    wpk[kh][1] = t2 | (t3 << 16);
    wt4e[kh] = t4;
    wt4o[kh] = t4 << 16;
  }
  const float b = bias[co];
  const int run = (qpi + 7) / 8;        // 25 pooled pixels per slot
  int q0 = slot * run;
  int q1 = min(qpi, q0 + run);
  ushort_t* yi = y + (size_t)img * qpi * Cout;
  uint8_t* ai = amax + (size_t)img * qpi * Cout;
  unsigned wv[6][3];  // window: 3 even-aligned bf16 pairs per row
  int prev_ho = -9;
  for (int q = q0; q < q1; ++q) {
    int wo = q % Wo, ho = q / Wo;
    int oy = ho * 2, ox = wo * 2;       // slab coords = conv coords + 2 pad
    if (ho != prev_ho || wo == 0) {
#pragma unroll
      for (int r = 0; r < 6; ++r)
#pragma unroll
        for (int c = 0; c < 3; ++c)
          wv[r][c] = *reinterpret_cast<const unsigned*>(&slab[oy + r][ox + c * 2]);
      prev_ho = ho;
    } else {
      // shift left by one pair, read the new pair
#pragma unroll
      for (int r = 0; r < 6; ++r) {
        wv[r][0] = wv[r][1];
        wv[r][1] = wv[r][2];
        wv[r][2] = *reinterpret_cast<const unsigned*>(&slab[oy + r][ox + 4]);
      }
    }
    // two partials per output (even/odd kh) -> 8 independent dot2 chains:
    // a single chain per output serialized 15 dependent v_dot2c (~5-cycle
    // RAW each — PMC showed 71% SQ_WAIT_INST_ANY); 8 chains interleave
    // past the latency
    // NOTE: an 8-chain even/odd-kh accumulator split was tried against the
    // 71% SQ_WAIT_INST_ANY reading and measured SLOWER (38/232 us vs
    // 34/206 at B=1024/8192) — the stall is not the acc RAW chain; the
    // 4-chain nested-dot2 form below is the measured optimum.
    // The alignbit cross-pairs are HOISTED ahead of the dot2 chains
    // (+12 VGPR at 28-register headroom) so no v_alignbit result is
    // consumed the cycle after it issues.
    unsigned oA[6][2];
#pragma unroll
    for (int r = 0; r < 6; ++r) {
      oA[r][0] = __builtin_amdgcn_alignbit(wv[r][1], wv[r][0], 16);
      oA[r][1] = __builtin_amdgcn_alignbit(wv[r][2], wv[r][1], 16);
    }
    float acc0 = b, acc1 = b, acc2 = b, acc3 = b;
#pragma unroll
    for (int kh = 0; kh < 5; ++kh) {
      acc0 = dot2bf(wv[kh][0], wpk[kh][0],
             dot2bf(wv[kh][1], wpk[kh][1],
             dot2bf(wv[kh][2], wt4e[kh], acc0)));
      acc1 = dot2bf(oA[kh][0], wpk[kh][0],
             dot2bf(oA[kh][1], wpk[kh][1],
             dot2bf(wv[kh][2], wt4o[kh], acc1)));
      acc2 = dot2bf(wv[kh + 1][0], wpk[kh][0],
             dot2bf(wv[kh + 1][1], wpk[kh][1],
             dot2bf(wv[kh + 1][2], wt4e[kh], acc2)));
      acc3 = dot2bf(oA[kh + 1][0], wpk[kh][0],
             dot2bf(oA[kh + 1][1], wpk[kh][1],
             dot2bf(wv[kh + 1][2], wt4o[kh], acc3)));
    }
    float vals[4] = {acc0, acc1, acc2, acc3};
    float best = -1.0f / 0.0f;
    int barg = 0;
#pragma unroll
    for (int pz = 0; pz < 4; ++pz) {
      float v = vals[pz] > 0.f ? vals[pz] : 0.f;
      if (v > best) { best = v; barg = pz; }
    }
    yi[(size_t)q * Cout + co] = f2bf(best);
    // 7 = dead window (liveness for the pooled-consumer backward)
    ai[(size_t)q * Cout + co] = (uint8_t)(best > 0.f ? barg : 7);
  }
}

void launch_transpose_bf16(const unsigned short* src, unsigned short* dst,
                           int R, int C, hipStream_t s) {
  dim3 grid((C + 31) / 32, (R + 31) / 32);
  hipLaunchKernelGGL(transpose_bf16_kernel, grid, dim3(256), 0, s, src, dst,
                     R, C);
}

static TransposeBatchArgs make_batch_args(const TransposeDesc* descs, int n) {
  TransposeBatchArgs a{};
  int total = 0;
  for (int i = 0; i < n && i < 4; ++i) {
    a.d[i] = descs[i];
    a.tile0[i] = total;
    total += ((descs[i].C + 31) / 32) * ((descs[i].R + 31) / 32);
  }
  a.n = n < 4 ? n : 4;
  a.total = total;
  return a;
}

void launch_transpose_bf16_batch(const TransposeDesc* descs, int n,
                                 hipStream_t s) {
  TransposeBatchArgs a = make_batch_args(descs, n);
  hipLaunchKernelGGL(transpose_bf16_batch_kernel, dim3(a.total), dim3(256), 0,
                     s, a);
}

void launch_transpose_bf16_batch_adv(const TransposeDesc* descs, int n,
                                     long* step_dev, float* lr_scale_dev,
                                     float lr0, float decay, int decay_steps,
                                     float inv_contrib, hipStream_t s) {
  TransposeBatchArgs a = make_batch_args(descs, n);
  a.do_advance = 1;
  a.step_dev = step_dev;
  a.lr_scale_dev = lr_scale_dev;
  a.lr0 = lr0;
  a.decay = decay;
  a.decay_steps = decay_steps < 1 ? 1 : decay_steps;
  a.inv_contrib = inv_contrib;
  hipLaunchKernelGGL(transpose_bf16_batch_kernel, dim3(a.total), dim3(256), 0,
                     s, a);
}

void launch_conv1_direct_fwd(const unsigned short* x, const unsigned short* w,
                             const float* bias, unsigned short* y,
                             uint8_t* amax, int NB, int H, int W, int Cout,
                             hipStream_t s) {
  long mpool = (long)NB * (H / 2) * (W / 2);
  hipLaunchKernelGGL(conv1_direct_fwd_kernel, dim3(NB), dim3(256), 0, s, x,
                     w, bias, y, amax, NB, H, W, Cout, (int)mpool);
}

// ---------------------------------------------------------------------------
// conv1 backward dW, direct VALU form (Cin == 1, K = 25 taps).
// dw[kh][kw][co] = sum_{n,h,w} x[n, h+kh-2, w+kw-2] * dact[n,h,w,co].
// MFMA loses at K=25 (M-util 25/32 and the A gather is per-element); here
// each lane owns one co and 25 fp32 accumulators; the x image is staged once
// per block as the same padded fp32 [32][32] slab the conv1 forward uses,
// and the 5x5 x-window walks each row with a 5-phase cyclic column buffer —
// 5 broadcast LDS reads + 25 v_fmac per pixel.  A block accumulates G
// images into registers and flushes ONCE via an LDS-reduced 800-value
// atomicAdd (the per-image flush is what made the old slab path atomics-
// bound below NB=2048).
__global__ __launch_bounds__(256)
void conv1_dw_direct_kernel(const ushort_t* x, const ushort_t* dact,
                            float* dw, int NB, int H, int W, int Cout,
                            int G) {
  // math on v_dot2c_f32_bf16: pixel PAIRS (c, c+1) reduce with one dot2
  // per tap — acc[t] += x[c+kw-2]*g_c + x[c+kw-1]*g_{c+1} where the x pair
  // is adjacent in the packed bf16 slab (even pairs = one b32 read, odd
  // pairs = one v_alignbit) and the g pair packs two raw dact loads.
  // 25 dot2 + 10 alignbit per 2 pixels vs 50 scalar FMAs.
  __shared__ ushort_t slab[32][32];
  __shared__ float red[25][32];
  const int tid = threadIdx.x;
  const int co = tid & 31;
  const int hw = tid >> 5;  // half-wave 0..7: rows hw, hw+8, ...
  float acc[25];
#pragma unroll
  for (int t = 0; t < 25; ++t) acc[t] = 0.f;

  const int img0 = blockIdx.x * G;
  for (int g = 0; g < G; ++g) {
    const int img = img0 + g;
    if (img >= NB) break;  // uniform across the block
    const ushort_t* xi = x + (size_t)img * H * W;
    for (int i = tid; i < 32 * 32; i += 256) {
      int xx = i % 32, yy = i / 32;
      int sy = yy - 2, sx = xx - 2;
      ushort_t v = 0;
      if (sy >= 0 && sy < H && sx >= 0 && sx < W) v = xi[(size_t)sy * W + sx];
      slab[yy][xx] = v;
    }
    __syncthreads();
    const ushort_t* di = dact + (size_t)img * H * W * Cout + co;
    if (co < Cout) {
      for (int r = hw; r < H; r += 8) {
        // window: 3 even-aligned x pairs per tap row, covering x cols
        // c-2..c+3 for the current pixel pair (c even)
        unsigned win[5][3];
#pragma unroll
        for (int kr = 0; kr < 5; ++kr) {  // c=0 pairs: slab cols (0, 2, 4)
          win[kr][0] = *reinterpret_cast<const unsigned*>(&slab[r + kr][0]);
          win[kr][1] = *reinterpret_cast<const unsigned*>(&slab[r + kr][2]);
          win[kr][2] = *reinterpret_cast<const unsigned*>(&slab[r + kr][4]);
        }
        const ushort_t* drow = di + (size_t)r * W * Cout;
        // 3-pair-deep dact prefetch (6 loads in flight, raw u16 until use)
        ushort_t gcur[6], gnxt[6];
#pragma unroll
        for (int p = 0; p < 6; ++p)
          gcur[p] = (p < W) ? drow[(size_t)p * Cout] : (ushort_t)0;
        for (int cc = 0; cc < 28; cc += 6) {  // 3 pixel pairs per group
#pragma unroll
          for (int p = 0; p < 6; ++p) {
            int cn = cc + 6 + p;
            gnxt[p] = (cn < W) ? drow[(size_t)cn * Cout] : (ushort_t)0;
          }
#pragma unroll
          for (int j = 0; j < 3; ++j) {
            int c = cc + j * 2;
            if (c >= W) break;
            unsigned gp = (unsigned)gcur[j * 2] |
                          ((unsigned)gcur[j * 2 + 1] << 16);
#pragma unroll
            for (int kh = 0; kh < 5; ++kh) {
              unsigned o0 = __builtin_amdgcn_alignbit(win[kh][1], win[kh][0], 16);
              unsigned o1 = __builtin_amdgcn_alignbit(win[kh][2], win[kh][1], 16);
              acc[kh * 5 + 0] = dot2bf(win[kh][0], gp, acc[kh * 5 + 0]);
              acc[kh * 5 + 1] = dot2bf(o0, gp, acc[kh * 5 + 1]);
              acc[kh * 5 + 2] = dot2bf(win[kh][1], gp, acc[kh * 5 + 2]);
              acc[kh * 5 + 3] = dot2bf(o1, gp, acc[kh * 5 + 3]);
              acc[kh * 5 + 4] = dot2bf(win[kh][2], gp, acc[kh * 5 + 4]);
            }
            // advance window by one pair; clamp the last read inside the
            // slab (c=26 would index col 32 — the value is discarded but
            // an out-of-allocation LDS read can return NaN bits)
            int nc = c + 6 <= 30 ? c + 6 : 30;
#pragma unroll
            for (int kr = 0; kr < 5; ++kr) {
              win[kr][0] = win[kr][1];
              win[kr][1] = win[kr][2];
              win[kr][2] = *reinterpret_cast<const unsigned*>(
                  &slab[r + kr][nc]);
            }
          }
#pragma unroll
          for (int p = 0; p < 6; ++p) gcur[p] = gnxt[p];
        }
      }
    }
    __syncthreads();  // slab is re-staged next image
  }

  // block reduction: 8 half-waves -> LDS, then one global flush
  for (int i = tid; i < 25 * 32; i += 256) red[i / 32][i % 32] = 0.f;
  __syncthreads();
  if (co < Cout)
#pragma unroll
    for (int t = 0; t < 25; ++t) atomicAdd(&red[t][co], acc[t]);
  __syncthreads();
  for (int i = tid; i < 25 * Cout; i += 256)
    atomicAdd(&dw[(size_t)(i / Cout) * Cout + (i % Cout)],
              red[i / Cout][i % Cout]);
}

// ---------------------------------------------------------------------------
// conv1 dW + db consuming the POOLED gradient directly (pool backward fused
// at the consumer).  dact1 = scatter(dxc, am1) is 75% structural zeros and
// was the largest intermediate of the backward (B*28*28*32); this kernel
// never materializes it: each lane decodes (dy_pooled, y_pooled, argmax)
// into the two dense-row dot2 pair streams the direct kernel walked, so
// the pool_bwd_scatter kernel AND the dact1 write+read round trip
// (51 MB @ B=1024, 410 MB @ 8192) disappear from the critical path.
// db[c] = sum of masked pooled dy (what the scatter kernel used to flush).
// Same slab/window/dot2 math as conv1_dw_direct_kernel above.
// ---------------------------------------------------------------------------
__global__ __launch_bounds__(256)
void conv1_dw_pooled_kernel(const ushort_t* x, const ushort_t* dyp,
                            const uint8_t* am, float* dw, float* db,
                            int NB, int H, int W, int Cout, int G) {
  __shared__ ushort_t slab[32][32];
  __shared__ float red[26][32];  // 25 taps + db row
  const int tid = threadIdx.x;
  const int co = tid & 31;
  const int hw = tid >> 5;  // half-wave 0..7: pooled rows hw, hw+8
  const int Hp = H / 2, Wp = W / 2;
  float acc[25];
#pragma unroll
  for (int t = 0; t < 25; ++t) acc[t] = 0.f;
  float dbacc = 0.f;

  const int img0 = blockIdx.x * G;
  for (int g = 0; g < G; ++g) {
    const int img = img0 + g;
    if (img >= NB) break;  // uniform across the block
    const ushort_t* xi = x + (size_t)img * H * W;
    for (int i = tid; i < 32 * 32; i += 256) {
      int xx = i % 32, yy = i / 32;
      int sy = yy - 2, sx = xx - 2;
      ushort_t v = 0;
      if (sy >= 0 && sy < H && sx >= 0 && sx < W) v = xi[(size_t)sy * W + sx];
      slab[yy][xx] = v;
    }
    __syncthreads();
    if (co < Cout) {
      const size_t ib = ((size_t)img * Hp) * Wp * Cout + co;
      // issue ALL of this lane's pooled loads up front (rows hw and hw+8):
      // 56 independent loads in flight -> ONE latency wall per image
      // instead of one per row (PMC: 55% of wave cycles were parked)
      ushort_t rdy[2][14];
      uint8_t ram[2][14];
#pragma unroll
      for (int rr = 0; rr < 2; ++rr) {
        int rq = hw + rr * 8;
        if (rq >= Hp) break;
#pragma unroll
        for (int qx = 0; qx < 14; ++qx) {
          size_t o = ib + ((size_t)rq * Wp + qx) * Cout;
          rdy[rr][qx] = dyp[o];
          ram[rr][qx] = am[o];
        }
      }
#pragma unroll
      for (int rr = 0; rr < 2; ++rr) {
        int rq = hw + rr * 8;
        if (rq >= Hp) break;
        // packed (bf16 value | argmax<<16) per qx, from the prefetched raws
        unsigned pk[14];
#pragma unroll
        for (int qx = 0; qx < 14; ++qx) {
          int pos = ram[rr][qx];  // 0..3 live, 7 dead (liveness in the byte)
          float gv = pos < 4 ? bf2f(rdy[rr][qx]) : 0.f;
          pk[qx] = (unsigned)f2bf(gv) | ((unsigned)pos << 16);
        }
#pragma unroll
        for (int ry = 0; ry < 2; ++ry) {
          const int r = 2 * rq + ry;
          unsigned win[5][3];
#pragma unroll
          for (int kr = 0; kr < 5; ++kr) {
            win[kr][0] = *reinterpret_cast<const unsigned*>(&slab[r + kr][0]);
            win[kr][1] = *reinterpret_cast<const unsigned*>(&slab[r + kr][2]);
            win[kr][2] = *reinterpret_cast<const unsigned*>(&slab[r + kr][4]);
          }
          for (int j = 0; j < 14; ++j) {  // dense pair c = 2j
            unsigned pkj = pk[j];
            unsigned gb = pkj & 0xffffu;
            int pos = (int)(pkj >> 16);
            unsigned gp = ((pos >> 1) == ry)
                              ? ((pos & 1) ? (gb << 16) : gb)
                              : 0u;
#pragma unroll
            for (int kh = 0; kh < 5; ++kh) {
              unsigned o0 = __builtin_amdgcn_alignbit(win[kh][1], win[kh][0], 16);
              unsigned o1 = __builtin_amdgcn_alignbit(win[kh][2], win[kh][1], 16);
              acc[kh * 5 + 0] = dot2bf(win[kh][0], gp, acc[kh * 5 + 0]);
              acc[kh * 5 + 1] = dot2bf(o0, gp, acc[kh * 5 + 1]);
              acc[kh * 5 + 2] = dot2bf(win[kh][1], gp, acc[kh * 5 + 2]);
              acc[kh * 5 + 3] = dot2bf(o1, gp, acc[kh * 5 + 3]);
              acc[kh * 5 + 4] = dot2bf(win[kh][2], gp, acc[kh * 5 + 4]);
            }
            int nc = 2 * j + 6 <= 30 ? 2 * j + 6 : 30;
#pragma unroll
            for (int kr = 0; kr < 5; ++kr) {
              win[kr][0] = win[kr][1];
              win[kr][1] = win[kr][2];
              win[kr][2] = *reinterpret_cast<const unsigned*>(&slab[r + kr][nc]);
            }
          }
          if (ry == 0) {
            // db once per pooled row (value independent of ry)
#pragma unroll
            for (int j = 0; j < 14; ++j) dbacc += bf2f((ushort_t)(pk[j] & 0xffffu));
          }
        }
      }
    }
    __syncthreads();  // slab re-staged next image
  }

  // block reduction: 8 half-waves -> LDS, then one global flush
  for (int i = tid; i < 26 * 32; i += 256) red[i / 32][i % 32] = 0.f;
  __syncthreads();
  if (co < Cout) {
#pragma unroll
    for (int t = 0; t < 25; ++t) atomicAdd(&red[t][co], acc[t]);
    atomicAdd(&red[25][co], dbacc);
  }
  __syncthreads();
  for (int i = tid; i < 25 * Cout; i += 256)
    atomicAdd(&dw[(size_t)(i / Cout) * Cout + (i % Cout)],
              red[i / Cout][i % Cout]);
  if (db && tid < Cout) atomicAdd(&db[tid], red[25][tid]);
}

void launch_conv1_dw_pooled(const unsigned short* x, const unsigned short* dyp,
                            const uint8_t* am, float* dw, float* db, int NB,
                            int H, int W, int Cout, hipStream_t s) {
  // G (images per block) trades parallelism against the 800-address
  // atomic flush each block pays: sweep (DMNIST_DW1_G) measured
  // 72.7/57.2/55.2/87.4 us at G=1/2/4/8 for B=1024 and 455/274/254 at
  // G=2/4/8 for B=8192
  int G = NB >= 4096 ? 8 : (NB >= 512 ? 4 : 1);
  if (const char* e = getenv("DMNIST_DW1_G")) G = atoi(e);  // flush sweep
  if (G < 1) G = 1;
  int blocks = (NB + G - 1) / G;
  hipLaunchKernelGGL(conv1_dw_pooled_kernel, dim3(blocks), dim3(256), 0, s,
                     x, dyp, am, dw, db, NB, H, W, Cout, G);
}

void launch_conv1_dw_direct(const unsigned short* x,
                            const unsigned short* dact, float* dw, int NB,
                            int H, int W, int Cout, hipStream_t s) {
  int G = NB >= 8192 ? 8 : (NB >= 2048 ? NB / 1024 : 1);
  int blocks = (NB + G - 1) / G;
  hipLaunchKernelGGL(conv1_dw_direct_kernel, dim3(blocks), dim3(256), 0, s,
                     x, dact, dw, NB, H, W, Cout, G);
}

// ---------------------------------------------------------------------------
// Maxpool 2x2 backward, GATHER form: one thread-granule per OUTPUT pixel
// (8 channels).  The scatter form above writes 4 separate 16B stores per
// window at a 2C-element stride — half-dense store streams that ran
// 2.6-3.5x above the traffic floor.  Here every thread writes exactly its
// own 16B of dact in layout order (fully dense, no read-for-ownership
// partial lines); the pooled dy/y/amax reads are 4x redundant but
// L2-broadcast (the 4 output pixels of a window read the same 16B).
// db accumulates from the pos==0 visitor only (each window counted once).
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256)
void pool_bwd_gather_kernel(const ushort_t* dy, const ushort_t* y,
                            const uint8_t* amax, ushort_t* dact, float* db,
                            int Mpool, int C, int H, int W, int Wo) {
  __shared__ float partial[64];
  if (threadIdx.x < 64) partial[threadIdx.x] = 0.f;
  __syncthreads();
  const unsigned CB8 = (unsigned)C / 8;
  const unsigned cb8_sh = (CB8 == 8) ? 3u : (CB8 == 4 ? 2u : 0u);
  const bool cb8_pow2 = (CB8 & (CB8 - 1)) == 0;
  const unsigned Ho = (unsigned)H / 2;
  unsigned total = (unsigned)Mpool * 4u * CB8;  // output granules
  unsigned stride = gridDim.x * blockDim.x;
  float local[8] = {0, 0, 0, 0, 0, 0, 0, 0};
  int mycol = -1;
  for (unsigned i = blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    unsigned cb, pix;
    if (cb8_pow2 && cb8_sh) { cb = i & (CB8 - 1); pix = i >> cb8_sh; }
    else { cb = i % CB8; pix = i / CB8; }
    int c0 = (int)cb * 8;
    unsigned w_ = pix % (unsigned)W;
    unsigned t2 = pix / (unsigned)W;
    unsigned h_ = t2 % (unsigned)H;
    unsigned n = t2 / (unsigned)H;
    int pos = (int)((h_ & 1u) * 2u + (w_ & 1u));
    size_t qb = (((size_t)n * Ho + (h_ >> 1)) * (unsigned)Wo + (w_ >> 1)) * C
                + c0;
    short8 dyv = *reinterpret_cast<const short8*>(dy + qb);
    short8 yv = *reinterpret_cast<const short8*>(y + qb);
    uint64_t am8;
    __builtin_memcpy(&am8, amax + qb, 8);
    short8 out;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      float g = bf2f((ushort_t)dyv[e]);
      if (!(bf2f((ushort_t)yv[e]) > 0.f)) g = 0.f;
      int am = (int)((am8 >> (8 * e)) & 0xff);
      out[e] = (am == pos) ? (short)f2bf(g) : (short)0;
      if (pos == 0) local[e] += g;  // one visitor per window counts db
    }
    *reinterpret_cast<short8*>(dact + (size_t)i * 8) = out;
    if (mycol < 0) mycol = c0;
  }
  if (mycol >= 0) {
#pragma unroll
    for (int e = 0; e < 8; ++e)
      atomicAdd(&partial[(mycol + e) & 63], local[e]);
  }
  __syncthreads();
  if (db && threadIdx.x < 64) {
    float v = partial[threadIdx.x];
    if (C >= 64) {
      if (v != 0.f) atomicAdd(&db[threadIdx.x], v);
    } else if ((int)threadIdx.x < C) {
      float v2 = partial[threadIdx.x + 32];
      if (v + v2 != 0.f) atomicAdd(&db[threadIdx.x], v + v2);
    }
  }
}

// ---------------------------------------------------------------------------
// Split-K forward epilogue: y[m,n] = [drop][relu](sum_s acc[s][m][n] +
// bias[n]) in bf16.  The slice sum runs in a FIXED order so the forward is
// bitwise-reproducible per (seed, offset) — the reason the split-K fwd
// writes per-slice planes instead of fp32 atomics.  Philox indexing
// matches the fused GEMM epilogue exactly: idx = m * N + n.
// ---------------------------------------------------------------------------
extern "C" __global__ __launch_bounds__(256)
void fwd_epilogue_kernel(const float* acc, const float* bias, ushort_t* y,
                         int M, int N, int slices, int relu, float p_keep,
                         uint64_t seed, uint64_t offset,
                         const long* offset_dev) {
  if (offset_dev) offset = (uint64_t)*offset_dev;
  size_t total = (size_t)M * N;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += stride) {
    float v = 0.f;
    for (int s = 0; s < slices; ++s) v += acc[(size_t)s * total + i];
    int n = (int)(i % (size_t)N);
    v += bias[n];
    if (relu) v = v > 0.f ? v : 0.f;
    if (p_keep < 1.0f) {
      float u = philox_uniform(seed, offset, (uint64_t)i);
      v = (u < p_keep) ? v / p_keep : 0.f;
    }
    y[i] = f2bf(v);
  }
}

void launch_fwd_epilogue(const float* acc, const float* bias,
                         unsigned short* y, int M, int N, int slices,
                         int relu, float p_keep, uint64_t seed,
                         uint64_t offset, const long* offset_dev,
                         hipStream_t s) {
  long total = (long)M * N;
  int blocks = cdivh(total, 256 * 4);
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(fwd_epilogue_kernel, dim3(blocks), dim3(256), 0, s, acc,
                     bias, y, M, N, slices, relu, p_keep, seed, offset,
                     offset_dev);
}
