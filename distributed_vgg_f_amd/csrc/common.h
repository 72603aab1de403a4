// Common device/host helpers for the MI355X (gfx950, CDNA4) kernels.
// Wave size is 64 everywhere; block sizes are multiples of 64.
#pragma once

#include <hip/hip_runtime.h>
#include <cstdint>

#define DEV __device__ __forceinline__

typedef __attribute__((ext_vector_type(2))) float f32x2;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;
typedef __attribute__((ext_vector_type(4))) short s16x4;
typedef __attribute__((ext_vector_type(8))) short s16x8;
typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4;

using u16 = unsigned short;
using u32 = unsigned int;
using i64 = long long;

// ---- bf16 <-> f32 -------------------------------------------------------
DEV float bf2f(u16 v) {
  union { u32 u; float f; } t;
  t.u = u32(v) << 16;
  return t.f;
}

// round-to-nearest-even f32 -> bf16 (matches hardware/torch casts)
DEV u16 f2bf(float f) {
  union { float f; u32 u; } t;
  t.f = f;
  u32 u = t.u;
  if ((u & 0x7fffffffu) > 0x7f800000u) return u16(0x7fc0);  // NaN
  u32 round = 0x7fffu + ((u >> 16) & 1u);
  return u16((u + round) >> 16);
}

DEV f32x2 bf2x2(u32 packed) {
  f32x2 r;
  r.x = bf2f(u16(packed & 0xffffu));
  r.y = bf2f(u16(packed >> 16));
  return r;
}

// ---- magic division -----------------------------------------------------
// Host-precomputed multiply-shift division (dividends < 2^28): replaces
// ~20-cycle v_rcp-based u32 divides in the gather hot path with one 64-bit
// multiply + shift.
struct MDiv {
  unsigned long long mul;
  int sh;
  unsigned int d;
  MDiv() : mul(1), sh(0), d(1) {}
  explicit MDiv(unsigned int d_) : d(d_) {
    int L = 0;
    while ((1u << L) < d_) ++L;
    sh = 32 + L;
    mul = ((1ull << sh) + d_ - 1) / d_;
  }
  DEV u32 div(u32 x) const { return u32(((unsigned long long)x * mul) >> sh); }
  DEV u32 mod(u32 x, u32 q) const { return x - q * d; }
};

// ---- grid helpers -------------------------------------------------------
// Bijective XCD-aware block remap (cdna_hip_programming.md section 5, T1):
// consecutive remapped ids stay on one XCD so neighbor tiles share its L2.
DEV u32 xcd_swizzle(u32 wgid, u32 nwg) {
  constexpr u32 NXCD = 8;
  if (nwg < 2 * NXCD) return wgid;
  u32 xcd = wgid % NXCD, idx = wgid / NXCD;
  u32 q = nwg / NXCD, r = nwg % NXCD;
  u32 base = (xcd < r) ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q;
  return base + idx;
}

__host__ __forceinline__ int ceil_div(long a, long b) {
  return int((a + b - 1) / b);
}

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e), " at ",     \
                  __FILE__, ":", __LINE__);                                 \
    }                                                                       \
  } while (0)
