// Ring direct convolution for Cin=64 stride-1 3x3 layers (K1/K2 fast
// path: VGG-16 conv1_2/conv2_1, ResNet layer1, and their dgrads).
//
// The implicit-GEMM conv path reads a VIRTUAL A panel of M x K elements —
// a KH*KW-fold re-read of x through L2/L3 (VGG-16 conv1_2 fwd: 14.7 GB
// virtual vs 1.6 GB of x; the layer measured ~1.35 ms, L3/HBM-bound on
// the expansion). Here BOTH operands are LDS-resident:
//   * the whole wT B panel (Kfull x 64 co = 73 KB for 3x3 Cin=64) is
//     staged ONCE per block — zero B traffic in the loop;
//   * the x tile lives in an NSS-slot ring: a block owns one
//     (n, 32-column segment) run and walks oh in TR=4-row tiles, staging
//     ONE new 4-row set per tile (the KH-1 halo rows stay resident, as
//     in wgrad_ring); every tap (kh, kw) reads the same slab at a
//     row/column offset. The fragment k-dim is ci (contiguous), so tap
//     reads are plain aligned ds_read_b128 — no shift windows.
// One barrier per TILE (not per k-step): the k-loop is pure LDS reads +
// MFMAs; the only VMEM in flight is the per-tile x set (counted vmcnt).
//
// The SAME kernel computes stride-1 dgrad: dx = conv(dy, wFlipT) with
// pad' = KH-1-pad (host supplies the flipped/transposed weight, cached
// per optimizer step).
//
// Geometry: 8 waves (512 threads) as 4(M) x 2(N); per-wave 32 output
// positions x 32 co; M-tile = 128 positions = 4 rows x 32 cols; BNC=64.
#pragma once

#include "common.h"
#include "gemm_tile.h"   // zero8, as_bf16x8, xcd_swizzle

namespace dvgg {

// EPI: 0 plain, 1 +bias, 2 +bias+ReLU, 3 ReLU
template <int EPI>
__global__ __launch_bounds__(512) void conv_ring_kernel(
    const u16* __restrict__ x,    // [N][H][W][64] NHWC (fwd) / dy (dgrad)
    const u16* __restrict__ wT,   // [Cout][Kfull = 9*64]
    const u16* __restrict__ bias, // [Cout] or null
    u16* __restrict__ y,          // [N][OH][OW][Cout]
    const u16* __restrict__ zpage, int Nn, int H, int W, int Cout,
    int OH, int OW, int OWp2, int pad, int runs_total, int LT) {
  constexpr int KH = 3, KW = 3, CIN = 64;
  constexpr int NI = 2;                  // 2 x 16-position fragments/wave
  constexpr int CT = 32;                 // output column segment
  constexpr int TR = 4;                  // output rows per tile (128 pos)
  constexpr int RH = TR + KH - 1;
  constexpr int S0 = (RH + TR - 1) / TR; // 2 window sets
  constexpr int NSS = S0 + 1;            // x ring slots (prefetch depth 1)
  constexpr int WWP = CT + 8;
  constexpr int BNC = 64;
  constexpr int KFULL = KH * KW * CIN;   // 576
  constexpr int KS = KFULL / 32;         // 18
  // x set: TR rows x WWP cols x 64 ci = 1280 chunks -> pad 1536 (3/thread)
  constexpr int CPR = WWP * (CIN / 8);   // 320 chunks per x row
  constexpr int ACH = (TR * CPR + 511) / 512 * 512;
  constexpr int ASI = ACH / 512;         // 3
  constexpr int SETE = ACH * 8;
  constexpr int BCH = KFULL / 8 * BNC;   // 4608 chunks (9/thread)
  constexpr int BSI = BCH / 512;
  constexpr int BE = BCH * 8;            // 73728 elems
  __shared__ u16 smem[NSS * SETE + BE];

  const int SEG = OWp2 / CT;
  const int co_tiles = Cout / BNC;
  u32 wg = xcd_swizzle(blockIdx.x, runs_total * co_tiles);
  const int run = int(wg) / co_tiles;
  const int co0 = int(wg) % co_tiles * BNC;
  const int n_r = run / SEG;
  const int c0 = (run % SEG) * CT;

  const int t = threadIdx.x;
  const int lane = t & 63, wave = t >> 6;
  const int wm = (wave >> 1) * 32;       // position sub-tile base
  const int wn = (wave & 1) * 32;
  const int l15 = lane & 15;
  const int kfrag = (lane >> 4) * 8;     // ci offset within a 32-k step

  // ---- staging descriptors (glds dest wave-uniform + lane*16)
  int a_r[ASI], a_iw[ASI], a_kof[ASI]; bool a_ok[ASI];
#pragma unroll
  for (int g = 0; g < ASI; ++g) {
    int c = g * 512 + t;
    a_r[g] = c / CPR;
    int rem = c % CPR;
    a_iw[g] = rem / (CIN / 8);
    a_kof[g] = (rem % (CIN / 8)) * 8;
    a_ok[g] = c < TR * CPR;
  }
  u16* const a_dst0 = &smem[(i64)(wave * 64) * 8];
  u16* const b_dst0 = &smem[NSS * SETE + (i64)(wave * 64) * 8];

#define CR_STAGE_X(slot, s_)                                                 \
  do {                                                                       \
    _Pragma("unroll")                                                        \
    for (int g = 0; g < ASI; ++g) {                                          \
      int ih_ = (s_) * TR + a_r[g] - pad;                                    \
      int iw_ = c0 + a_iw[g] - pad;                                          \
      const u16* src_ = (a_ok[g] && (unsigned)ih_ < (unsigned)H &&           \
                         (unsigned)iw_ < (unsigned)W)                        \
          ? x + (((i64)n_r * H + ih_) * W + iw_) * CIN + a_kof[g]            \
          : zpage;                                                           \
      __builtin_amdgcn_global_load_lds(                                      \
          (const __attribute__((address_space(1))) unsigned int*)src_,       \
          (__attribute__((address_space(3))) unsigned int*)                  \
              (a_dst0 + (i64)(slot) * SETE + (i64)g * 512 * 8),              \
          16, 0, 0);                                                         \
    }                                                                        \
  } while (0)

  // ---- prologue: the WHOLE B panel once + x sets for tiles 0 and 1.
  // B chunk c = (kq, co): B image [KFULL/8][BNC][8]; source wT row co.
#pragma unroll
  for (int g = 0; g < BSI; ++g) {
    int c = g * 512 + t;
    int kq = c / BNC, co = c % BNC;
    const u16* src_ = wT + (i64)(co0 + co) * KFULL + kq * 8;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)src_,
        (__attribute__((address_space(3))) unsigned int*)
            (b_dst0 + (i64)g * 512 * 8),
        16, 0, 0);
  }
#pragma unroll
  for (int s = 0; s < S0; ++s) CR_STAGE_X(s, s);
  int sfree = S0;
  int sbase = 0;

  f32x4 acc[NI][2];
  const u16* const Bs_ = &smem[NSS * SETE];
  for (int tile = 0; tile < LT; ++tile) {
    const int oh0 = tile * TR;
    // tile's window + the B panel are in LDS (the set staged last tile
    // had a whole k-loop to land — the drain is cheap)
    asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();        // also fences x slot reuse
    // stage tile+1's new x set (= set tile+2) under this tile's compute
    // (sets past the run read the zero page)
    CR_STAGE_X(sfree, tile + S0);
    sfree = sfree + 1 == NSS ? 0 : sfree + 1;

#pragma unroll
    for (int i = 0; i < NI; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
    for (int ks = 0; ks < KS; ++ks) {
      const int tap = ks >> 1;           // CIN=64: 2 k-steps per tap
      const int ci0 = (ks & 1) * 32;
      const int kh = tap / KW, kw = tap % KW;
      bf16x8 bfr[2];
#pragma unroll
      for (int j = 0; j < 2; ++j)
        bfr[j] = as_bf16x8(*reinterpret_cast<const s16x8*>(
            Bs_ + ((i64)((ks * 32 + kfrag) >> 3) * BNC + wn + j * 16 + l15) *
                      8));
      bf16x8 afr[NI];
#pragma unroll
      for (int i = 0; i < NI; ++i) {
        int p = wm + i * 16 + l15;
        int q = p / CT + kh;             // window row (tile-relative)
        int slot = sbase + q / TR;
        if (slot >= NSS) slot -= NSS;
        afr[i] = as_bf16x8(*reinterpret_cast<const s16x8*>(
            &smem[0] + (i64)slot * SETE +
            ((i64)(q % TR) * WWP + p % CT + kw) * CIN + ci0 + kfrag));
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < NI; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i], bfr[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    sbase = sbase + 1 == NSS ? 0 : sbase + 1;

    // ---- epilogue: this tile's 128 x 64 outputs (bias/ReLU fused)
    float bvj[2];
#pragma unroll
    for (int j = 0; j < 2; ++j) {
      int col = co0 + wn + j * 16 + l15;
      bvj[j] = (EPI == 1 || EPI == 2) ? bf2f(bias[col]) : 0.f;
    }
#pragma unroll
    for (int i = 0; i < NI; ++i) {
      int pbase = wm + i * 16 + (lane >> 4) * 4;
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        int col = co0 + wn + j * 16 + l15;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int p = pbase + r;
          int oh = oh0 + p / CT, ow = c0 + p % CT;
          if (oh >= OH || ow >= OW) continue;
          float v = acc[i][j][r] + bvj[j];
          if (EPI == 2 || EPI == 3) v = v > 0.f ? v : 0.f;
          y[(((i64)n_r * OH + oh) * OW + ow) * Cout + col] = f2bf(v);
        }
      }
    }
  }
}

}  // namespace dvgg
