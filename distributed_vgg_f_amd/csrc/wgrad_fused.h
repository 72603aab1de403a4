// Tap-fused 3x3 stride-1 wgrad (SURVEY.md K3, round-2 structural fix).
//
// The round-1 wgrad lowered dw[(kh,kw,ci)][co] as a GEMM with
// M = KH*KW*Cin, so the dy B-panel was re-read M/BM times (7-9x; PMC:
// 21% MFMA / 63% wait, 23% of the VGG-F step). Here the GEMM M-dim is
// ci ONLY:
//   * kh enters as an LDS ROW OFFSET into a staged x slab that covers the
//     k-tile's output rows plus KH-1 halo rows — all three kh taps read
//     the same slab;
//   * kw enters as a REGISTER WINDOW: per (ci-frag, kh) the kernel reads
//     12 contiguous bf16 (one ds_read_b128 + one ds_read_b64) and derives
//     the kw=0/2 fragments by dword re-selection (free) and kw=1 by four
//     v_alignbit (VALU, hidden under MFMA);
//   * each wave holds 9 accumulator sets (one per tap), 32ci x 32co each.
// dy is then read Cin/64 times per co-tile instead of KH*KW*Cin/256 —
// conv3: 245 MB -> 109 MB of B traffic; per-k-tile arithmetic intensity
// 87 -> ~230 FLOP/B.
//
// Geometry: 2(m) x WAVES_N waves, 64 ci x (32*WAVES_N) co block, BK=32
// k-positions per tile where k = (n, oh, ow) over the OWp2-padded output
// (OWp2 = OW padded to a multiple of 8 with OWp2 % 32 == 0 or
// 32 % OWp2 == 0, so a k-tile is R = 32/CT full rows (CT = OWp2 < 32) or
// a 32-column segment of one row (CT = 32)).
//
// LDS (one __shared__ arena, glds-staged, double-buffered):
//   A slab [rr = R+2][ci 64][CTW = CT+8]: CTW is an odd multiple of 4
//   dwords, so the 16-lane ds_read_b128 fragment groups (ci varies per
//   lane) land on 16 distinct bank quads — conflict-free without an XOR
//   swizzle, while the glds destination stays lane-linear.
//   B tile [kofs 4][co][8]: fragment k-elems contiguous per co; 4-dword
//   co stride -> conflict-free b128 reads.
#pragma once

#include "common.h"
#include "gemm_tile.h"   // zero8, as_bf16x8, xcd_swizzle

namespace dvgg {

// CT = k-tile column extent; BMC = 64 ci rows; BNC = 32*WAVES_N co cols.
// WAVES_M=2 -> 32 ci per wave (2 A-frags; 3x3 fits 9x2x2 accumulators),
// WAVES_M=4 -> 16 ci per wave (1 A-frag; 5x5 fits 25x1x2).
template <int CT, int WAVES_M, int WAVES_N, int KH, int KW>
__global__ __launch_bounds__((WAVES_M * WAVES_N) * 64) void wgrad_fused3_kernel(
    const u16* __restrict__ xT,    // [Cin][N][Hp][Wp], Wp = OWp2 + 8
    const u16* __restrict__ dyT,   // [Cout][N][OH][OWp2], zero col pad
    float* __restrict__ c32,       // [slices][KH*KW*Cin][Cout] slabs
    const u16* __restrict__ zpage, int Cin, int Cout, int N, int Hp, int Wp,
    int OH, int OWp2, int tiles_k, int slices) {
  constexpr int NW = WAVES_M * WAVES_N;    // waves per block
  constexpr int THREADS_W = NW * 64;
  constexpr int NI = WAVES_M == 2 ? 2 : 1; // A fragments per wave
  constexpr int BMC = WAVES_M * 16 * NI;   // = 64 ci rows per block
  constexpr int BNC = 32 * WAVES_N;
  static_assert(KW <= 5, "kw window covers 12 staged elements");
  constexpr int R = 32 / CT;               // output rows per k-tile
  constexpr int RH = R + KH - 1;           // staged slab rows
  constexpr int CTW = CT + 8;              // slab cols (halo + align slack)
  constexpr int CPR = CTW / 8;             // 16B chunks per slab row
  constexpr int ACH_RAW = RH * BMC * CPR;  // real A chunks
  constexpr int ACH = (ACH_RAW + THREADS_W - 1) / THREADS_W * THREADS_W;
  constexpr int ASLAB = ACH * 8;           // elems (incl. slack)
  constexpr int BCH = 4 * BNC;             // B chunks (one per k-frag x co)
  constexpr int BCHP = (BCH + THREADS_W - 1) / THREADS_W * THREADS_W;
  constexpr int BSLAB = BCHP * 8;
  // 3-buffer ring + counted vmcnt (cf. gemm_glds3_kernel): this kernel
  // runs 1 block/CU (195-197 VGPR), so a per-tile vmcnt(0) drain has no
  // occupancy to hide behind — stages are issued two tiles ahead and the
  // per-tile wait leaves the newest stage in flight.
  __shared__ u16 smem[3][ASLAB + BSLAB];

  const int tiles_m = Cin / BMC, tiles_n = Cout / BNC;
  u32 wg = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int ci0 = int(wg) / tiles_n * BMC;
  const int co0 = int(wg) % tiles_n * BNC;
  // split-K slice over whole k-tiles
  const int span = (tiles_k + slices - 1) / slices;
  const int t_lo = blockIdx.y * span;
  const int t_hi = min(tiles_k, t_lo + span);
  if (t_lo >= t_hi) return;

  const int t = threadIdx.x;
  const int lane = t & 63, wave = t >> 6;
  const int wm = (wave / WAVES_N) * (16 * NI);  // ci sub-tile base
  const int wn = (wave % WAVES_N) * 32;    // co sub-tile base
  const int l15 = lane & 15;
  const int kfrag = (lane >> 4) * 8;       // k offset of this lane's frags

  // tile decomposition over the padded per-image k-space
  const int SEG = (CT == 32) ? (OWp2 / 32) : 1;     // col segments per row
  const int OHK = (OH + R - 1) / R * R;
  const int TPI = (CT == 32) ? OH * SEG : OHK / R;  // k-tiles per image

  // --- per-slot staging descriptors. glds dest must be WAVE-UNIFORM
  // (hardware adds lane*16 itself); the per-lane part lives entirely in
  // the SOURCE address. chunk c = slot base (wave-uniform) + lane.
  constexpr int AGI = ACH / THREADS_W;     // A glds per thread
  constexpr int BGI = BCHP / THREADS_W;
  int a_rr[AGI]; i64 a_cibase[AGI]; int a_col[AGI]; bool a_ok[AGI];
  u16* a_dst[AGI];
#pragma unroll
  for (int g = 0; g < AGI; ++g) {
    int c = g * THREADS_W + t;             // this lane's chunk id
    int rr = c / (BMC * CPR), rem = c % (BMC * CPR);
    int ci = rem / CPR, kof = rem % CPR;
    a_rr[g] = rr;
    a_cibase[g] = ((i64)(ci0 + ci) * N) * Hp * (i64)Wp;
    a_col[g] = kof * 8;
    a_ok[g] = c < ACH_RAW;
    a_dst[g] = &smem[0][(i64)(g * THREADS_W + wave * 64) * 8];
  }
  int b_kofs[BGI]; i64 b_cobase[BGI]; bool b_ok[BGI]; u16* b_dst[BGI];
#pragma unroll
  for (int g = 0; g < BGI; ++g) {
    int c = g * THREADS_W + t;
    int kofs = c / BNC, co = c % BNC;
    b_kofs[g] = kofs;
    b_cobase[g] = ((i64)(co0 + co) * N) * OH * (i64)OWp2;
    b_ok[g] = c < BCH;
    b_dst[g] = &smem[0][ASLAB + (i64)(g * THREADS_W + wave * 64) * 8];
  }

#define WGF_STAGE(pbuf, tile_id)                                             \
  do {                                                                       \
    int n_ = (tile_id) / TPI, r_ = (tile_id) % TPI;                          \
    int oh0_ = (CT == 32) ? r_ / SEG : r_ * R;                               \
    int c0_ = (CT == 32) ? (r_ % SEG) * 32 : 0;                              \
    _Pragma("unroll")                                                        \
    for (int g = 0; g < AGI; ++g) {                                          \
      int row_ = oh0_ + a_rr[g];                                             \
      const u16* src_ = (a_ok[g] && row_ < Hp)                               \
          ? xT + a_cibase[g] + ((i64)n_ * Hp + row_) * Wp + c0_ + a_col[g]   \
          : zpage;                                                           \
      __builtin_amdgcn_global_load_lds(                                      \
          (const __attribute__((address_space(1))) unsigned int*)src_,       \
          (__attribute__((address_space(3))) unsigned int*)                  \
              (a_dst[g] + (pbuf) * (ASLAB + BSLAB)),                         \
          16, 0, 0);                                                         \
    }                                                                        \
    _Pragma("unroll")                                                        \
    for (int g = 0; g < BGI; ++g) {                                          \
      int k8_ = b_kofs[g] * 8;                                               \
      int oh_ = oh0_ + k8_ / CT;                                             \
      int colk_ = c0_ + k8_ % CT;                                            \
      const u16* src_ = (b_ok[g] && oh_ < OH)                                \
          ? dyT + b_cobase[g] + ((i64)n_ * OH + oh_) * OWp2 + colk_          \
          : zpage;                                                           \
      __builtin_amdgcn_global_load_lds(                                      \
          (const __attribute__((address_space(1))) unsigned int*)src_,       \
          (__attribute__((address_space(3))) unsigned int*)                  \
              (b_dst[g] + (pbuf) * (ASLAB + BSLAB)),                         \
          16, 0, 0);                                                         \
    }                                                                        \
  } while (0)

  // per-wave output: (16*NI) ci x 32 co per tap
  static_assert(WAVES_N == 2 || WAVES_N == 4, "WAVES_N in {2,4}");
  constexpr int NJ = 2;  // 32 co per wave = 2 x 16-col fragments
  f32x4 acc[KH * KW][NI][NJ];
#pragma unroll
  for (int tp = 0; tp < KH * KW; ++tp)
#pragma unroll
    for (int i = 0; i < NI; ++i)
#pragma unroll
      for (int j = 0; j < NJ; ++j)
        acc[tp][i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nt = t_hi - t_lo;
  constexpr int SPT = AGI + BGI;           // glds per thread per stage
  WGF_STAGE(0, t_lo);
  if (nt > 1) WGF_STAGE(1, t_lo + 1);

  int p = 0;
  for (int tile = 0; tile < nt; ++tile) {
    // keep exactly the newest stage in flight across the wait; the ring
    // write target (tile+2)%3 was last read at tile-1, one barrier back
    if (tile + 1 < nt)
      asm volatile("s_waitcnt vmcnt(%0)" ::"i"(SPT) : "memory");
    else
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    __builtin_amdgcn_s_barrier();
    const u16* As_ = &smem[p][0];
    const u16* Bs_ = As_ + ASLAB;
    // B fragments: [kofs][co][8], co = wn + j*16 + l15
    bf16x8 bfr[NJ];
#pragma unroll
    for (int j = 0; j < NJ; ++j)
      bfr[j] = as_bf16x8(*reinterpret_cast<const s16x8*>(
          Bs_ + ((i64)(kfrag >> 3) * BNC + wn + j * 16 + l15) * 8));
    const int rA = kfrag / CT;             // slab row of this k-frag
    const int cA = kfrag % CT;             // col within slab row
#pragma unroll
    for (int kh = 0; kh < KH; ++kh) {
      u32 d[NI][6];
#pragma unroll
      for (int i = 0; i < NI; ++i) {
        const u16* base = As_ +
            ((i64)(rA + kh) * BMC + wm + i * 16 + l15) * CTW + cA;
        *reinterpret_cast<s16x8*>(&d[i][0]) =
            *reinterpret_cast<const s16x8*>(base);        // ds_read_b128
        *reinterpret_cast<s16x4*>(&d[i][4]) =
            *reinterpret_cast<const s16x4*>(base + 8);    // ds_read_b64
      }
      if (kh == 0) {
        // issue tile+2's stage under the MFMA stream (ring slot is free)
        asm volatile("" ::: "memory");
        if (tile + 2 < nt) WGF_STAGE((tile + 2) % 3, t_lo + tile + 2);
      }
#pragma unroll
      for (int kw = 0; kw < KW; ++kw) {
        bf16x8 afr[NI];
#pragma unroll
        for (int i = 0; i < NI; ++i) {
          u32 w[4];
          if (kw % 2 == 0) {       // even shift: dword re-select, free
#pragma unroll
            for (int q = 0; q < 4; ++q) w[q] = d[i][kw / 2 + q];
          } else {                 // odd shift: 16-bit funnel
#pragma unroll
            for (int q = 0; q < 4; ++q)
              w[q] = __builtin_amdgcn_alignbit(d[i][kw / 2 + 1 + q],
                                               d[i][kw / 2 + q], 16);
          }
          afr[i] = as_bf16x8(*reinterpret_cast<const s16x8*>(&w[0]));
        }
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < NI; ++i)
#pragma unroll
          for (int j = 0; j < NJ; ++j)
            acc[kh * KW + kw][i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                afr[i], bfr[j], acc[kh * KW + kw][i][j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
    }
    p = p + 1 == 3 ? 0 : p + 1;
  }
#undef WGF_STAGE

  // epilogue: plain fp32 stores into THIS SLICE's slab (atomics measured
  // 0.32 ms/step on VGG-F — every slice re-adding the whole dw output
  // serialized at the L2 atomic units; slabs + the summing finalize move
  // sk*|dw| bytes twice instead, ~3x cheaper). Every (tile, slice) block
  // writes its whole 64ci x BNC region, so the slab needs NO zero-fill.
  float* slab = c32 + (i64)blockIdx.y * ((i64)KH * KW * Cin) * Cout;
#pragma unroll
  for (int tp = 0; tp < KH * KW; ++tp) {
#pragma unroll
    for (int i = 0; i < NI; ++i) {
      int row_base = tp * Cin + ci0 + wm + i * 16 + (lane >> 4) * 4;
#pragma unroll
      for (int j = 0; j < NJ; ++j) {
        int col = co0 + wn + j * 16 + l15;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          slab[(i64)(row_base + r) * Cout + col] = acc[tp][i][j][r];
      }
    }
  }
}



// ------------------------------------------------------------------ ring
// Rolling-window variant: within one (n, c0) COLUMN RUN the x rows of
// consecutive k-tiles overlap by KH-1 rows; the plain kernel re-stages
// the whole RH-row slab every tile (3x A traffic for CT=32, 2x for
// CT=16). Here tiles walk oh within a run and each tile stages exactly
// ONE new R-row "set" into an NSS-slot LDS ring; the KH-1 halo rows stay
// resident. Run transitions batch-stage the next run's S0+1 sets during
// the second-to-last tile, keeping the counted-vmcnt discipline at two
// static values (steady vmcnt(2); vmcnt(S0+3) across the seam).
// Slices are whole runs; host guarantees LT >= 3 tiles per run.
template <int CT, int WAVES_M, int WAVES_N, int KH, int KW>
__global__ __launch_bounds__((WAVES_M * WAVES_N) * 64) void wgrad_ring_kernel(
    const u16* __restrict__ xT,    // [Cin][N][Hp][Wp], Wp = OWp2 + 8
    const u16* __restrict__ dyT,   // [Cout][N][OH][OWp2], zero col pad
    float* __restrict__ c32,       // [slices][KH*KW*Cin][Cout] slabs
    const u16* __restrict__ zpage, int Cin, int Cout, int N, int Hp, int Wp,
    int OH, int OWp2, int runs_total, int LT, int slices) {
  constexpr int NW = WAVES_M * WAVES_N;
  constexpr int THREADS_W = NW * 64;
  static_assert(THREADS_W == 512, "ring kernel assumes 512 threads");
  constexpr int NI = WAVES_M == 2 ? 2 : 1;
  constexpr int BMC = WAVES_M * 16 * NI;   // 64
  constexpr int BNC = 32 * WAVES_N;
  constexpr int R = 32 / CT;               // rows per tile / per set
  constexpr int RH = R + KH - 1;           // live window rows
  constexpr int S0 = (RH + R - 1) / R;     // window sets
  // Pipeline lookahead in TILES. LK=3 (deeper A/B rings, steady
  // vmcnt(4), seam at LT-LK staging LK next-run tiles) was MEASURED
  // WORSE: +6% on the 3x3 ct32 variant (906 vs 855 us — more in-flight
  // sets pressure L2 without covering more useful latency) and +15% on
  // ct16 (the 80->112 KB LDS growth halved its 2-workgroups/CU
  // occupancy). The counts below stay generic in LK for re-evaluation.
  constexpr int LK = 2;
  constexpr int NSS = 2 * S0 + 2 * (LK - 1);  // A ring slots (seam safe)
  constexpr int CTW = CT + 8;
  constexpr int CPR = CTW / 8;
  constexpr int SETC = R * BMC * CPR;      // real chunks per set (<=512)
  static_assert(SETC <= 512, "one glds per thread per set");
  constexpr int SETE = 512 * 8;            // padded set elems (8 KiB)
  constexpr int BTC = 4 * BNC;             // real B chunks per tile
  constexpr int BTE = 512 * 8;             // padded B tile elems (8 KiB)
  constexpr int NB = 2 * LK;               // B ring slots
  __shared__ u16 smem[NSS * SETE + NB * BTE];

  const int tiles_m = Cin / BMC, tiles_n = Cout / BNC;
  u32 wg = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int ci0 = int(wg) / tiles_n * BMC;
  const int co0 = int(wg) % tiles_n * BNC;
  const int SEG = (CT == 32) ? (OWp2 / 32) : 1;
  // slice = whole runs
  const int rspan = (runs_total + slices - 1) / slices;
  const int r_lo = blockIdx.y * rspan;
  const int r_hi = min(runs_total, r_lo + rspan);
  if (r_lo >= r_hi) return;

  const int t = threadIdx.x;
  const int lane = t & 63, wave = t >> 6;
  const int wm = (wave / WAVES_N) * (16 * NI);
  const int wn = (wave % WAVES_N) * 32;
  const int l15 = lane & 15;
  const int kfrag = (lane >> 4) * 8;

  // ---- staging descriptors (one glds per thread per A set / B tile)
  // A chunk c = (r, ci, kof) within a set; dest wave-uniform
  const int a_r = t / (BMC * CPR);
  const int a_ci = (t % (BMC * CPR)) / CPR;
  const int a_kof = (t % CPR) * 8;
  const bool a_ok = t < SETC;
  const i64 a_cib = ((i64)(ci0 + a_ci) * N) * Hp * (i64)Wp;
  u16* const a_dst0 = &smem[(i64)(wave * 64) * 8];
  const int b_kofs = t / BNC;
  const int b_co = t % BNC;
  const bool b_ok = t < BTC;               // BNC=64: half the threads idle
  const i64 b_cob = ((i64)(co0 + b_co) * N) * OH * (i64)OWp2;
  u16* const b_dst0 = &smem[NSS * SETE + (i64)(wave * 64) * 8];

  // stage A set s (run-local) of run (n_, c0_) into ring slot 'slot'
#define WGR_STAGE_A(slot, n_, c0_, s_)                                       \
  do {                                                                       \
    int row_ = (s_) * R + a_r;                                               \
    const u16* src_ = (a_ok && row_ < Hp)                                    \
        ? xT + a_cib + ((i64)(n_) * Hp + row_) * Wp + (c0_) + a_kof          \
        : zpage;                                                             \
    __builtin_amdgcn_global_load_lds(                                        \
        (const __attribute__((address_space(1))) unsigned int*)src_,         \
        (__attribute__((address_space(3))) unsigned int*)                    \
            (a_dst0 + (i64)(slot) * SETE),                                   \
        16, 0, 0);                                                           \
  } while (0)
#define WGR_STAGE_B(bslot, n_, c0_, tile_)                                   \
  do {                                                                       \
    int k8_ = b_kofs * 8;                                                    \
    int oh_ = (tile_) * R + k8_ / CT;                                        \
    const u16* src_ = (b_ok && oh_ < OH)                                     \
        ? dyT + b_cob + ((i64)(n_) * OH + oh_) * OWp2 + (c0_) + k8_ % CT     \
        : zpage;                                                             \
    __builtin_amdgcn_global_load_lds(                                        \
        (const __attribute__((address_space(1))) unsigned int*)src_,         \
        (__attribute__((address_space(3))) unsigned int*)                    \
            (b_dst0 + (i64)(bslot) * BTE),                                   \
        16, 0, 0);                                                           \
  } while (0)

  f32x4 acc[KH * KW][NI][2];
#pragma unroll
  for (int tp = 0; tp < KH * KW; ++tp)
#pragma unroll
    for (int i = 0; i < NI; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) acc[tp][i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  // run-local state; slot counters advance monotonically mod ring size
  int run = r_lo;
  int n_r = run / SEG, c0_r = (run % SEG) * 32;
  int sbase = 0;       // ring slot of the current tile's FIRST window set
  int bbase = 0;
  // prologue of the first run. ISSUE ORDER MATTERS: vmcnt counts in
  // issue order, so tile0's stages go first and tile1's (A top set + B)
  // last — the steady vmcnt(2) then leaves exactly tile1's pair in
  // flight while guaranteeing tile0's window.
#pragma unroll
  for (int s = 0; s < S0; ++s) WGR_STAGE_A(s % NSS, n_r, c0_r, s);
  WGR_STAGE_B(0, n_r, c0_r, 0);
#pragma unroll
  for (int lk = 1; lk < LK; ++lk) {        // tiles 1..LK-1 of the run
    WGR_STAGE_A((S0 + lk - 1) % NSS, n_r, c0_r, S0 + lk - 1);
    WGR_STAGE_B(lk, n_r, c0_r, lk);
  }
  int sfree = (S0 + LK - 1) % NSS;  // next free A slot
  int bfree = LK;

  for (; run < r_hi; ++run) {
    if (run > r_lo) { n_r = run / SEG; c0_r = (run % SEG) * 32; }
    const bool last_run = run + 1 == r_hi;
    const int nn = (run + 1 < r_hi) ? (run + 1) / SEG : 0;
    const int nc0 = (run + 1 < r_hi) ? ((run + 1) % SEG) * 32 : 0;
    for (int tile = 0; tile < LT; ++tile) {
      // ---- wait + barrier: current tile's window and B tile are in LDS
      // younger-than-this-tile's-pair load counts (in-order retirement):
      // steady = the LK-1 staged-ahead pairs; behind==1/0 tails shrink by
      // 2 each; a pending seam batch adds its S0 + 2*LK - 1 loads.
      const int behind = LT - 1 - tile;
      if (last_run) {
        if (behind == 0)
          asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
        else if (LK == 3 && behind == 1)
          asm volatile("s_waitcnt vmcnt(2)" ::: "memory");
        else
          asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2 * (LK - 1)) : "memory");
      } else {
        if (behind == 0)
          asm volatile("s_waitcnt vmcnt(%0)" ::"i"(S0 + 2 * LK - 1)
                       : "memory");
        else if (LK == 3 && behind == 1)
          asm volatile("s_waitcnt vmcnt(%0)" ::"i"(S0 + 7) : "memory");
        else
          asm volatile("s_waitcnt vmcnt(%0)" ::"i"(2 * (LK - 1)) : "memory");
      }
      __builtin_amdgcn_s_barrier();

      const u16* Bs_ = &smem[NSS * SETE + (i64)bbase * BTE];
      bf16x8 bfr[2];
#pragma unroll
      for (int j = 0; j < 2; ++j)
        bfr[j] = as_bf16x8(*reinterpret_cast<const s16x8*>(
            Bs_ + ((i64)(kfrag >> 3) * BNC + wn + j * 16 + l15) * 8));
      const int rA = kfrag / CT;
      const int cA = kfrag % CT;
#pragma unroll
      for (int kh = 0; kh < KH; ++kh) {
        u32 d[NI][6];
#pragma unroll
        for (int i = 0; i < NI; ++i) {
          int q = rA + kh;                       // window row 0..RH-1
          int slot = sbase + q / R;
          if (slot >= NSS) slot -= NSS;
          const u16* base = &smem[0] + (i64)slot * SETE +
              ((i64)(q % R) * BMC + wm + i * 16 + l15) * CTW + cA;
          *reinterpret_cast<s16x8*>(&d[i][0]) =
              *reinterpret_cast<const s16x8*>(base);
          *reinterpret_cast<s16x4*>(&d[i][4]) =
              *reinterpret_cast<const s16x4*>(base + 8);
        }
        if (kh == 0) {
          asm volatile("" ::: "memory");
          // ---- issue next stages under the MFMAs
          if (tile + LK < LT) {                 // regular: tile+LK's set
            WGR_STAGE_A(sfree, n_r, c0_r, tile + LK + S0 - 1);
            sfree = sfree + 1 == NSS ? 0 : sfree + 1;
            WGR_STAGE_B(bfree, n_r, c0_r, tile + LK);
            bfree = bfree + 1 == NB ? 0 : bfree + 1;
          } else if (tile == LT - LK && !last_run) {
            // seam batch: next run's prologue in the prologue's issue
            // order (tile0' window + B0', then tile1' top + B1')
#pragma unroll
            for (int s = 0; s < S0; ++s) {
              int sl = sfree + s;
              while (sl >= NSS) sl -= NSS;
              WGR_STAGE_A(sl, nn, nc0, s);
            }
            WGR_STAGE_B(bfree, nn, nc0, 0);
            bfree = bfree + 1 == NB ? 0 : bfree + 1;
#pragma unroll
            for (int lk = 1; lk < LK; ++lk) {
              int sl = sfree + S0 + lk - 1;
              while (sl >= NSS) sl -= NSS;
              WGR_STAGE_A(sl, nn, nc0, S0 + lk - 1);
              WGR_STAGE_B(bfree, nn, nc0, lk);
              bfree = bfree + 1 == NB ? 0 : bfree + 1;
            }
            sfree = sfree + S0 + LK - 1;
            while (sfree >= NSS) sfree -= NSS;
          }
        }
#pragma unroll
        for (int kw = 0; kw < KW; ++kw) {
          bf16x8 afr[NI];
#pragma unroll
          for (int i = 0; i < NI; ++i) {
            u32 w[4];
            if (kw % 2 == 0) {
#pragma unroll
              for (int q = 0; q < 4; ++q) w[q] = d[i][kw / 2 + q];
            } else {
#pragma unroll
              for (int q = 0; q < 4; ++q)
                w[q] = __builtin_amdgcn_alignbit(d[i][kw / 2 + 1 + q],
                                                 d[i][kw / 2 + q], 16);
            }
            afr[i] = as_bf16x8(*reinterpret_cast<const s16x8*>(&w[0]));
          }
          __builtin_amdgcn_s_setprio(1);
#pragma unroll
          for (int i = 0; i < NI; ++i)
#pragma unroll
            for (int j = 0; j < 2; ++j)
              acc[kh * KW + kw][i][j] =
                  __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                      afr[i], bfr[j], acc[kh * KW + kw][i][j], 0, 0, 0);
          __builtin_amdgcn_s_setprio(0);
        }
      }
      sbase = sbase + 1 == NSS ? 0 : sbase + 1;   // window advances 1 set
      bbase = bbase + 1 == NB ? 0 : bbase + 1;
    }
    // window of the next run starts at its prologue's first slot: the
    // in-run advance moved sbase by LT-1... recompute from sfree:
    // next run's sets occupy [sfree - (S0+1), sfree) -> first = that
    if (!last_run) {
      sbase = sfree - (S0 + LK - 1);
      while (sbase < 0) sbase += NSS;
    }
  }
#undef WGR_STAGE_A
#undef WGR_STAGE_B

  float* slab = c32 + (i64)blockIdx.y * ((i64)KH * KW * Cin) * Cout;
#pragma unroll
  for (int tp = 0; tp < KH * KW; ++tp) {
#pragma unroll
    for (int i = 0; i < NI; ++i) {
      int row_base = tp * Cin + ci0 + wm + i * 16 + (lane >> 4) * 4;
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        int col = co0 + wn + j * 16 + l15;
#pragma unroll
        for (int r = 0; r < 4; ++r)
          slab[(i64)(row_base + r) * Cout + col] = acc[tp][i][j][r];
      }
    }
  }
}

}  // namespace dvgg
