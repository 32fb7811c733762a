// Common device helpers for the distributedmnist_amd CDNA4 (gfx950) kernels.
// bf16 scalar/vector types, philox4x32-10 RNG, bounds helpers.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <stdint.h>

typedef __attribute__((ext_vector_type(8))) short short8;     // 8 x bf16 (4 VGPR)
typedef __attribute__((ext_vector_type(4))) float f32x4;      // MFMA 16x16 acc
typedef unsigned short ushort_t;

#define DEV __device__ __forceinline__

DEV float bf2f(ushort_t u) {
  union { uint32_t i; float f; } v;
  v.i = uint32_t(u) << 16;
  return v.f;
}

DEV ushort_t f2bf(float f) {
  union { float f; uint32_t i; } v;
  v.f = f;
  // round-to-nearest-even
  uint32_t lsb = (v.i >> 16) & 1;
  uint32_t rounded = v.i + 0x7FFFu + lsb;
  if ((v.i & 0x7F800000u) == 0x7F800000u) rounded = v.i;  // inf/nan passthrough
  return ushort_t(rounded >> 16);
}

// ---------------------------------------------------------------------------
// philox4x32-10 (counter-based; same stream for a given (seed, offset, idx))
// ---------------------------------------------------------------------------
struct Philox4 {
  uint32_t x, y, z, w;
};

DEV uint32_t mulhilo(uint32_t a, uint32_t b, uint32_t* hi) {
  uint64_t p = uint64_t(a) * uint64_t(b);
  *hi = uint32_t(p >> 32);
  return uint32_t(p);
}

DEV Philox4 philox4x32(uint64_t seed, uint64_t offset, uint64_t idx) {
  uint32_t k0 = uint32_t(seed), k1 = uint32_t(seed >> 32);
  uint32_t c0 = uint32_t(idx), c1 = uint32_t(idx >> 32);
  uint32_t c2 = uint32_t(offset), c3 = uint32_t(offset >> 32);
  const uint32_t M0 = 0xD2511F53u, M1 = 0xCD9E8D57u;
  const uint32_t W0 = 0x9E3779B9u, W1 = 0xBB67AE85u;
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    uint32_t h0, h1;
    uint32_t l0 = mulhilo(M0, c0, &h0);
    uint32_t l1 = mulhilo(M1, c2, &h1);
    uint32_t n0 = h1 ^ c1 ^ k0;
    uint32_t n1 = l1;
    uint32_t n2 = h0 ^ c3 ^ k1;
    uint32_t n3 = l0;
    c0 = n0; c1 = n1; c2 = n2; c3 = n3;
    k0 += W0; k1 += W1;
  }
  return {c0, c1, c2, c3};
}

// uniform in [0,1) from one lane of philox
DEV float philox_uniform(uint64_t seed, uint64_t offset, uint64_t idx) {
  Philox4 p = philox4x32(seed, offset, idx);
  return float(p.x) * (1.0f / 4294967296.0f);
}

DEV int cdiv(int a, int b) { return (a + b - 1) / b; }

// ---- glds + ds_read_b64_tr_b16 staging primitives (gfx950) ----------------
// Semantics pinned empirically: profiles/r01_tr16_probe.md.  The glds LDS
// destination is wave-uniform base + lane*16 (hardware); tr16 gives each
// lane of a 16-lane group img[kq + j][n0 + (l&15)] (j = 0..3) from a
// K-MAJOR LDS image when lane g of the group addresses
// &img[kq + (g>>2)][n0 + 4*(g&3)].  Keep these OUT of function templates:
// a target builtin inside a template makes hipcc's host pass silently skip
// emitting the kernel's device stub.
DEV void glds16(const ushort_t* src, ushort_t* dst) {
  __builtin_amdgcn_global_load_lds(src, dst, 16, 0, 0);
}

// un-waited transpose read: issue-only, drain with pack_wait before use
DEV uint2 tr16_issue(unsigned a) {
  uint2 d;
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0"
               : "=&v"(d) : "v"(a) : "memory");
  return d;
}

// pack two 4-element halves into an MFMA fragment; the wait is tied to the
// packed value so no MFMA reading it can be scheduled before the drain
// (repeat waits after the first are ~free: the counter is already 0)
DEV short8 pack_wait(uint2 lo, uint2 hi) {
  union { unsigned u[4]; short8 s; } r;
  r.u[0] = lo.x; r.u[1] = lo.y; r.u[2] = hi.x; r.u[3] = hi.y;
  asm volatile("s_waitcnt lgkmcnt(0)" : "+v"(r.s));
  return r.s;
}

// packed bf16 dot product: c += a.x*b.x + a.y*b.y (v_dot2c_f32_bf16,
// 2 MACs/cycle/lane — 2x the fp32 FMA rate)
typedef short bf16x2_t __attribute__((__vector_size__(2 * sizeof(short))));
DEV float dot2bf(unsigned a, unsigned b, float c) {
  union U { unsigned u; bf16x2_t v; };
  U ua, ub;
  ua.u = a; ub.u = b;
  return __builtin_amdgcn_fdot2_f32_bf16(ua.v, ub.v, c, false);
}
