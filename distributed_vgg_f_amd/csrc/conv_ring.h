// Ring direct convolution for small-Cin stride-1 layers (K1/K2 fast path).
//
// The implicit-GEMM conv path reads a VIRTUAL A panel of M x K elements —
// a KH*KW-fold re-read of x through L2/L3 (VGG-16 conv1_2 fwd: 14.7 GB
// virtual vs 1.6 GB of x; the layer measured ~1.35 ms, i.e. L3/HBM-bound
// on the expansion). Here the x tile is LDS-RESIDENT across the whole
// K-loop: a block owns one (n, 32-column segment) run and walks oh in
// TR-row tiles, staging ONE new TR-row x set per tile into an NSS-slot
// ring (the KH-1 halo rows stay resident, as in wgrad_ring); every tap
// (kh, kw) reads the same slab at a row/column offset. Unlike the wgrad
// case the fragment k-dim is ci (contiguous), so tap reads are plain
// aligned ds_read_b128 — no shift windows.
//
// The SAME kernel computes stride-1 dgrad: dx = conv(dy, wFlipT) with
// pad' = KH-1-pad (host supplies the flipped/transposed weight, cached
// per optimizer step).
//
// Geometry: 8 waves (512 threads) as WAVES_M=4 (M) x WAVES_N (N);
// per-wave 32 output positions x 32 co. M-tile = 128 positions =
// TR = 128/CT rows of a CT=32 column segment. B (wT [Cout][Kfull]) is
// staged per 32-k step into a 3-slot ring with counted vmcnt (the wT
// re-read per M-tile is L2-resident by construction: Kfull*BNC*2 bytes).
#pragma once

#include "common.h"
#include "gemm_tile.h"   // zero8, as_bf16x8, xcd_swizzle

namespace dvgg {

// EPI: 0 plain, 1 +bias, 2 +bias+ReLU, 3 ReLU
template <int WAVES_N, int KH, int KW, int EPI>
__global__ __launch_bounds__(512) void conv_ring_kernel(
    const u16* __restrict__ x,    // [N][H][W][Cin] NHWC (fwd) / dy (dgrad)
    const u16* __restrict__ wT,   // [Cout][KH*KW*Cin]
    const u16* __restrict__ bias, // [Cout] or null
    u16* __restrict__ y,          // [N][OH][OW][Cout]
    const u16* __restrict__ zpage, int Nn, int H, int W, int Cin, int Cout,
    int OH, int OW, int OWp2, int pad, int runs_total, int LT) {
  constexpr int WAVES_M = 4;
  constexpr int NI = 2;                  // 2 x 16-position fragments/wave
  constexpr int CT = 32;                 // output column segment
  constexpr int TR = 4;                  // output rows per tile (128 pos)
  constexpr int RH = TR + KH - 1;        // live x rows
  constexpr int S0 = (RH + TR - 1) / TR; // window sets (2 for KH<=5)
  constexpr int NSS = S0 + 3;            // x ring slots (4 in use + 1)
  constexpr int WWP = CT + 8;            // staged columns (halo KW-1<=8)
  constexpr int BNC = 32 * WAVES_N;
  // x set: TR rows x WWP cols x Cin elems, padded to 512-chunk multiple
  // (Cin <= 64 gate keeps this within LDS)
  constexpr int CIMAX = 64;
  constexpr int ACH_RAW_MAX = TR * WWP * CIMAX / 8;        // 1280
  constexpr int ACH = (ACH_RAW_MAX + 511) / 512 * 512;     // 1536
  constexpr int ASI = ACH / 512;         // glds per thread per x set (3)
  constexpr int SETE = ACH * 8;          // elems per ring slot
  constexpr int BTE = 4 * BNC * 8;       // B tile elems (one 32-k step)
  constexpr int NB = 3;                  // B ring slots
  __shared__ u16 smem[NSS * SETE + NB * BTE];

  const int SEG = OWp2 / CT;
  const int co_tiles = Cout / BNC;
  u32 wg = xcd_swizzle(blockIdx.x, runs_total * co_tiles);
  const int run = int(wg) / co_tiles;
  const int co0 = int(wg) % co_tiles * BNC;
  const int n_r = run / SEG;
  const int c0 = (run % SEG) * CT;
  const int Kfull = KH * KW * Cin;
  const int KS = Kfull / 32;             // 32-k steps (Cin % 32 == 0)

  const int t = threadIdx.x;
  const int lane = t & 63, wave = t >> 6;
  const int wm = (wave / WAVES_N) * 32;  // position sub-tile base
  const int wn = (wave % WAVES_N) * 32;
  const int l15 = lane & 15;
  const int kfrag = (lane >> 4) * 8;     // ci offset within the 32-k step

  // ---- staging descriptors
  // x set chunk c = (r, iw, kof): per-row Cin elems, rows TR per set
  const int cpr = WWP * (Cin / 8);       // chunks per x row
  int a_r[ASI], a_iw[ASI], a_kof[ASI]; bool a_ok[ASI];
#pragma unroll
  for (int g = 0; g < ASI; ++g) {
    int c = g * 512 + t;
    a_r[g] = c / cpr;
    int rem = c % cpr;
    a_iw[g] = rem / (Cin / 8);
    a_kof[g] = (rem % (Cin / 8)) * 8;
    a_ok[g] = c < TR * cpr;
  }
  u16* const a_dst0 = &smem[(i64)(wave * 64) * 8];
  // B chunk: (kofs, co) over [4][BNC][8]; BNC=64 leaves half threads idle
  const int b_kofs = t / BNC;
  const int b_co = t % BNC;
  const bool b_ok = t < 4 * BNC;
  const u16* const b_src0 = wT + (i64)(co0 + b_co) * Kfull;
  u16* const b_dst0 = &smem[NSS * SETE + (i64)(wave * 64) * 8];

  // stage x set s (covers input rows s*TR - pad + r) into ring slot
#define CR_STAGE_X(slot, s_)                                                 \
  do {                                                                       \
    _Pragma("unroll")                                                        \
    for (int g = 0; g < ASI; ++g) {                                          \
      int ih_ = (s_) * TR + a_r[g] - pad;                                    \
      int iw_ = c0 + a_iw[g] - pad;                                          \
      const u16* src_ = (a_ok[g] && (unsigned)ih_ < (unsigned)H &&           \
                         (unsigned)iw_ < (unsigned)W)                        \
          ? x + (((i64)n_r * H + ih_) * W + iw_) * Cin + a_kof[g]            \
          : zpage;                                                           \
      __builtin_amdgcn_global_load_lds(                                      \
          (const __attribute__((address_space(1))) unsigned int*)src_,       \
          (__attribute__((address_space(3))) unsigned int*)                  \
              (a_dst0 + (i64)(slot) * SETE + (i64)g * 512 * 8),              \
          16, 0, 0);                                                         \
    }                                                                        \
  } while (0)
#define CR_STAGE_B(bslot, ks_)                                               \
  do {                                                                       \
    const u16* src_ = b_ok ? b_src0 + (ks_) * 32 + b_kofs * 8 : zpage;       \
    __builtin_amdgcn_global_load_lds(                                        \
        (const __attribute__((address_space(1))) unsigned int*)src_,         \
        (__attribute__((address_space(3))) unsigned int*)                    \
            (b_dst0 + (i64)(bslot) * BTE),                                   \
        16, 0, 0);                                                           \
  } while (0)

  f32x4 acc[NI][2];
  // prologue: x sets for tile0's window + tile1's new set; B steps 0, 1
  // (issue order matters: the counted waits below index from the newest)
#pragma unroll
  for (int s = 0; s < S0; ++s) CR_STAGE_X(s, s);
  CR_STAGE_B(0, 0);
  CR_STAGE_X(S0, S0);
  CR_STAGE_B(1, 1);
  int sfree = (S0 + 1) % NSS;
  int sbase = 0;
  int bbase = 0, bfree = 2;

  for (int tile = 0; tile < LT; ++tile) {
    const int oh0 = tile * TR;
#pragma unroll
    for (int i = 0; i < NI; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};
    for (int ks = 0; ks < KS; ++ks) {
      // counted waits, indexed from the newest in-flight stages; one
      // barrier per k-step bounds wave skew (B ring is 3 slots).
      // x is staged UNCONDITIONALLY at ks 0 (past-the-run sets read the
      // zero page) so the ks 1/2 counts stay uniform; the only special
      // case is the final k-step, which drains everything.
      if (tile == LT - 1 && ks == KS - 1)
        asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
      else if (ks == 1 || ks == 2)
        asm volatile("s_waitcnt vmcnt(%0)" ::"i"(1 + ASI) : "memory");
      else
        asm volatile("s_waitcnt vmcnt(1)" ::: "memory");
      __builtin_amdgcn_s_barrier();
      // ---- issue next stages (B two k-steps ahead; x set for tile+2
      // once per tile at ks 0)
      if (ks + 2 < KS || tile + 1 < LT) {
        int nks = ks + 2 >= KS ? ks + 2 - KS : ks + 2;
        CR_STAGE_B(bfree, nks);
        bfree = bfree + 1 == NB ? 0 : bfree + 1;
      }
      if (ks == 0) {
        CR_STAGE_X(sfree, tile + 2 + S0 - 1);
        sfree = sfree + 1 == NSS ? 0 : sfree + 1;
      }
      // ---- fragments
      const int tap = ks * 32 / Cin;     // Cin % 32 == 0
      const int ci0 = ks * 32 % Cin;
      const int kh = tap / KW, kw = tap % KW;
      const u16* Bs_ = &smem[NSS * SETE + (i64)bbase * BTE];
      bf16x8 bfr[2];
#pragma unroll
      for (int j = 0; j < 2; ++j)
        bfr[j] = as_bf16x8(*reinterpret_cast<const s16x8*>(
            Bs_ + ((i64)(kfrag >> 3) * BNC + wn + j * 16 + l15) * 8));
      bf16x8 afr[NI];
#pragma unroll
      for (int i = 0; i < NI; ++i) {
        int p = wm + i * 16 + l15;       // position within the tile
        int q = p / CT + kh;             // window row (tile-relative)
        int slot = sbase + q / TR;
        if (slot >= NSS) slot -= NSS;
        int r_in = q % TR;
        int iwl = p % CT + kw;           // column within the slab window
        afr[i] = as_bf16x8(*reinterpret_cast<const s16x8*>(
            &smem[0] + (i64)slot * SETE +
            ((i64)r_in * WWP + iwl) * Cin + ci0 + kfrag));
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < NI; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[i], bfr[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      bbase = bbase + 1 == NB ? 0 : bbase + 1;
    }
    sbase = sbase + 1 == NSS ? 0 : sbase + 1;

    // ---- epilogue: this tile's 128 x BNC outputs (bias/ReLU fused)
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
