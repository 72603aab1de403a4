// MFMA bf16 GEMM core for gfx950 — the single engine behind the FC layers
//  and all implicit-GEMM convolutions (K1/K2/K3): the
// "A"/"B" operands are gather functors, so conv fwd/dgrad/wgrad reuse one
// tile pipeline with different address functions.
//
// v2 geometry: 128 x (32*NFRAG) block tile (NFRAG=4 -> 128x128, NFRAG=2 ->
// 128x64 for Cout=64 layers), BK=64 k-step, 256 threads = 4 waves in a 2x2
// grid, each wave a 64 x (16*NFRAG) sub-tile of v_mfma_f32_16x16x32_bf16
// fragments, fp32 accumulation. ONE barrier per k-tile: double-buffered LDS
// where writing buf[p] at tile t is fenced from tile t-2's readers by tile
// t-1's barrier; next tile's global loads issue right after the barrier so
// HBM latency hides under the 32 MFMAs (T14 issue-early / write-late,
// cdna_hip_programming.md Guideline 15).
//
// LDS rows are [row][BK] bf16 with +8 row pad (144 B stride: 16 consecutive
// rows hit distinct bank groups for ds_read_b128 — Guideline 4).
//
// Gather functors expose prep(row) -> Row (hoists the row-side integer
// divisions out of the K loop) and load8(Row, k) -> 8 bf16 (zero where out
// of range; k is a multiple of 8).
#pragma once

#include "common.h"

namespace dvgg {

constexpr int BM = 128, BK = 32;
constexpr int LDS_PAD = 8;                 // bf16 elements
constexpr int LDS_STRIDE = BK + LDS_PAD;   // 40 elems = 80 B rows
constexpr int THREADS = 256;
constexpr int BN = 128;                    // default tile N (NFRAG=4)

DEV bf16x8 as_bf16x8(s16x8 v) {
  union { s16x8 s; bf16x8 b; } u;
  u.s = v;
  return u.b;
}

DEV s16x8 zero8() { return s16x8{0, 0, 0, 0, 0, 0, 0, 0}; }

// ---------------------------------------------------------------- gathers

struct PlainGather {
  const u16* p;
  int rows, cols, ld;  // ld multiple of 8 (16B-aligned rows)
  struct Row { const u16* rp; bool valid; };
  DEV Row prep(int m) const { return {p + (i64)m * ld, m < rows}; }
  DEV s16x8 load8(const Row& r, int k) const {
    if (r.valid && k < cols)
      return *reinterpret_cast<const s16x8*>(r.rp + k);
    return zero8();
  }
  DEV const u16* ptr8(const Row& r, int k, const u16* zpage) const {
    return (r.valid && k < cols) ? r.rp + k : zpage;
  }
  int kcontig() const { return cols; }  // host: dense K run for g8 gate
};

// which gathers can stage via global_load_lds (single 16B load or a zero
// page per chunk; funnel/scalar gathers cannot)
template <class G> struct GldsOk { static constexpr bool value = false; };
template <> struct GldsOk<PlainGather> { static constexpr bool value = true; };

// Implicit-GEMM forward A: virtual A[m=(n,oh,ow)][k=(kh,kw,ci)] = x NHWC.
// Requires Cin % 8 == 0 (chunks stay inside one filter tap) — Cin<8 first
// layers are channel-padded to 8 by the launcher. All index divisions are
// host-precomputed magic multiplies (MDiv).
struct ConvFwdGather {
  const u16* x;
  int M, H, W, Cin, stride, pad;
  MDiv dOHOW, dOW, dCin, dKW;
  struct Row { const u16* nrow; int ihb, iwb; bool valid; };
  DEV Row prep(int m) const {
    u32 n = dOHOW.div(m), r = dOHOW.mod(m, n);
    u32 oh = dOW.div(r), ow = dOW.mod(r, oh);
    return {x + (i64)n * H * W * Cin, int(oh) * stride - pad,
            int(ow) * stride - pad, m < M};
  }
  DEV s16x8 load8(const Row& rs, int k) const {
    u32 f = dCin.div(k), c0 = dCin.mod(k, f);
    u32 kh = dKW.div(f), kw = dKW.mod(f, kh);
    int ih = rs.ihb + int(kh), iw = rs.iwb + int(kw);
    if (rs.valid && (unsigned)ih < (unsigned)H && (unsigned)iw < (unsigned)W)
      return *reinterpret_cast<const s16x8*>(rs.nrow +
                                             ((i64)ih * W + iw) * Cin + c0);
    return zero8();
  }
  DEV const u16* ptr8(const Row& rs, int k, const u16* zpage) const {
    u32 f = dCin.div(k), c0 = dCin.mod(k, f);
    u32 kh = dKW.div(f), kw = dKW.mod(f, kh);
    int ih = rs.ihb + int(kh), iw = rs.iwb + int(kw);
    if (rs.valid && (unsigned)ih < (unsigned)H && (unsigned)iw < (unsigned)W)
      return rs.nrow + ((i64)ih * W + iw) * Cin + c0;
    return zpage;
  }
  int kcontig() const { return Cin; }   // host: K run inside one tap
};
template <> struct GldsOk<ConvFwdGather> { static constexpr bool value = true; };

// Implicit-GEMM dgrad A: virtual A[m=(n,ih,iw)][k=(kh,kw,co)] =
// dy[n][(ih+pad-kh)/s][(iw+pad-kw)/s][co]. Requires Cout % 8 == 0.
struct ConvDgradGather {
  const u16* dy;
  int M, H, W, Cout, OH, OW, stride, pad;
  MDiv dHW, dW_, dCout, dKW;
  struct Row { const u16* nrow; int ihp, iwp; bool valid; };
  DEV Row prep(int m) const {
    u32 n = dHW.div(m), r = dHW.mod(m, n);
    u32 ih = dW_.div(r), iw = dW_.mod(r, ih);
    return {dy + (i64)n * OH * OW * Cout, int(ih) + pad, int(iw) + pad,
            m < M};
  }
  DEV s16x8 load8(const Row& rs, int k) const {
    u32 f = dCout.div(k), c0 = dCout.mod(k, f);
    u32 kh = dKW.div(f), kw = dKW.mod(f, kh);
    int ohs = rs.ihp - int(kh), ows = rs.iwp - int(kw);
    int oh, ow;
    bool ok = rs.valid && ohs >= 0 && ows >= 0;
    if (stride == 1) {          // VGG path: no division at all
      oh = ohs; ow = ows;
    } else {
      oh = ohs / 2; ow = ows / 2;  // zoo uses stride 2 otherwise
      ok = ok && (ohs & 1) == 0 && (ows & 1) == 0;
    }
    ok = ok && oh < OH && ow < OW;
    if (ok)
      return *reinterpret_cast<const s16x8*>(rs.nrow +
                                             ((i64)oh * OW + ow) * Cout + c0);
    return zero8();
  }
  DEV const u16* ptr8(const Row& rs, int k, const u16* zpage) const {
    u32 f = dCout.div(k), c0 = dCout.mod(k, f);
    u32 kh = dKW.div(f), kw = dKW.mod(f, kh);
    int ohs = rs.ihp - int(kh), ows = rs.iwp - int(kw);
    int oh, ow;
    bool ok = rs.valid && ohs >= 0 && ows >= 0;
    if (stride == 1) { oh = ohs; ow = ows; }
    else { oh = ohs / 2; ow = ows / 2;
           ok = ok && (ohs & 1) == 0 && (ows & 1) == 0; }
    ok = ok && oh < OH && ow < OW;
    return ok ? rs.nrow + ((i64)oh * OW + ow) * Cout + c0 : zpage;
  }
  int kcontig() const { return Cout; }  // host: K run inside one tap
};
template <> struct GldsOk<ConvDgradGather> { static constexpr bool value = true; };

// Wgrad A (Cin % 8 == 0): virtual A[m=(kh,kw,ci)][k=(n,oh,ow-padded)] over
// xTpad [Cin][N][H+2p][Wp] (halo-padded, zero-filled; +8 elem slack). The
// k-space pads OW to OWp (mult of 8) so chunks never cross rows.
// phl (phase-plane mode, stride 2/4 small-Cin wgrad): xT holds
// [Cin][s][s] stride-phase planes of x (s = 1<<phl), each [N][Hp][Wp];
// ih = s*ihq + ph, iw = s*iwq + pw. A row (kh,kw,ci) then reads plane
// (ci, kh&(s-1), kw&(s-1)) at row offsets (kh>>phl, kw>>phl) — a STRIDE-1
// gather, so the b128+alignbit funnel path applies to strided convs
// (VGG-F conv1 s=4). phl=0 is the classic [Cin][N][Hp][Wp] layout.
struct WgradXGather {
  const u16* xT;
  int Mrows, Cin, N, Hp, Wp, OH, OWp, stride, Klim;
  MDiv dCin, dKW, dOHOWp, dOWp;
  int phl;
  struct Row { const u16* cib; int kh, kw; bool valid; };
  DEV Row prep(int m) const {
    u32 f = dCin.div(m), ci = dCin.mod(m, f);
    u32 kh = dKW.div(f), kw = dKW.mod(f, kh);
    u32 s1 = (1u << phl) - 1u;
    u32 plane = ((ci << phl | (kh & s1)) << phl) | (kw & s1);
    return {xT + (i64)plane * N * Hp * Wp, int(kh >> phl), int(kw >> phl),
            m < Mrows};
  }
  DEV s16x8 load8(const Row& rs, int k) const {
    if (!rs.valid || k >= Klim) return zero8();
    u32 n = dOHOWp.div(k), r = dOHOWp.mod(k, n);
    u32 oh = dOWp.div(r), ow = dOWp.mod(r, oh);
    s16x8 v;
    if (stride == 1 && int(dKW.d - 1) >> phl < 8) {
      // The row base (cib + ...*Wp) is 16B-aligned and ow is a multiple of
      // 8, so the only misalignment is +kw elements (kw < 8). Read the
      // aligned 32B window with TWO b128 loads (the kernel is VMEM-issue
      // bound; 5 dword loads cost 2.5x the TA slots) and extract 8 elems
      // with alignbit. Halo is pre-zeroed; buffer has +8 elem slack.
      const u16* A = rs.cib + ((i64)n * Hp + (oh + rs.kh)) * Wp + ow;
      s16x8 lo = *reinterpret_cast<const s16x8*>(A);
      s16x8 hi = *reinterpret_cast<const s16x8*>(A + 8);
      u32 w[8];
      *reinterpret_cast<s16x8*>(&w[0]) = lo;
      *reinterpret_cast<s16x8*>(&w[4]) = hi;
      u32* o = reinterpret_cast<u32*>(&v);
      int kw = rs.kw;
      u32 r = u32(kw & 1) * 16u;
      switch (kw >> 1) {  // row-constant: at most 3 targets per wave
        case 0:
          o[0] = __builtin_amdgcn_alignbit(w[1], w[0], r);
          o[1] = __builtin_amdgcn_alignbit(w[2], w[1], r);
          o[2] = __builtin_amdgcn_alignbit(w[3], w[2], r);
          o[3] = __builtin_amdgcn_alignbit(w[4], w[3], r);
          break;
        case 1:
          o[0] = __builtin_amdgcn_alignbit(w[2], w[1], r);
          o[1] = __builtin_amdgcn_alignbit(w[3], w[2], r);
          o[2] = __builtin_amdgcn_alignbit(w[4], w[3], r);
          o[3] = __builtin_amdgcn_alignbit(w[5], w[4], r);
          break;
        case 2:
          o[0] = __builtin_amdgcn_alignbit(w[3], w[2], r);
          o[1] = __builtin_amdgcn_alignbit(w[4], w[3], r);
          o[2] = __builtin_amdgcn_alignbit(w[5], w[4], r);
          o[3] = __builtin_amdgcn_alignbit(w[6], w[5], r);
          break;
        default:
          o[0] = __builtin_amdgcn_alignbit(w[4], w[3], r);
          o[1] = __builtin_amdgcn_alignbit(w[5], w[4], r);
          o[2] = __builtin_amdgcn_alignbit(w[6], w[5], r);
          o[3] = __builtin_amdgcn_alignbit(w[7], w[6], r);
          break;
      }
      return v;
    }
    const u16* base = rs.cib + ((i64)n * Hp + (int(oh) * stride + rs.kh)) * Wp +
                      (int(ow) * stride + rs.kw);
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = short(base[(i64)j * stride]);
    return v;
  }
};

// Wgrad A for Cin % 8 != 0: scalar gather straight from NHWC x (padding via
// per-element range predicates; padded-ow k's are zeroed by the B side).
struct WgradXDirectGather {
  const u16* x;
  int Mrows, Cin, N, H, W, OH, OWp, stride, pad, Klim;
  MDiv dCin, dKW, dOHOWp, dOWp;
  struct Row { const u16* cib; int kh, kw; bool valid; };
  DEV Row prep(int m) const {
    u32 f = dCin.div(m), ci = dCin.mod(m, f);
    u32 kh = dKW.div(f), kw = dKW.mod(f, kh);
    return {x + ci, int(kh), int(kw), m < Mrows};
  }
  DEV s16x8 load8(const Row& rs, int k) const {
    s16x8 v = zero8();
    if (!rs.valid || k >= Klim) return v;
    u32 n = dOHOWp.div(k), r = dOHOWp.mod(k, n);
    u32 oh = dOWp.div(r), ow0 = dOWp.mod(r, oh);
    int ih = int(oh) * stride - pad + rs.kh;
    if ((unsigned)ih >= (unsigned)H) return v;
    const u16* row = rs.cib + (((i64)n * H + ih) * W) * Cin;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int iw = (int(ow0) + j) * stride - pad + rs.kw;
      if ((unsigned)iw < (unsigned)W) v[j] = short(row[(i64)iw * Cin]);
    }
    return v;
  }
};

// glds-capable wgrad gathers (stride==1, Cin % 128 == 0: one filter tap
// per 128-row M-tile). The reduction is substituted to ow' = ow + kw - pad
// so the A side reads x at ALIGNED columns (x stored at column iw+8 in
// xTg [Cin][N][H+2p][Wg]); the per-tap column shift is baked into KW
// pre-shifted zero-padded copies of the transposed dy
// (dyTS [KW][Cout][N][OH][OWq]: copy kw holds dy[ow] at column
// ow + kw - pad + 8). The kernel passes the block-uniform tap to B.
struct WgradXgGather {
  const u16* xT;           // [Cin][N][Hp][Wg], column offset +8, zero halo
  int Mrows, Cin, N, Hp, Wg, OH, Wk, Klim;   // Wk = ceil8(W): k-space width
  MDiv dCin, dKW, dOHWk, dWk;
  struct Row { const u16* cib; int kh; bool valid; };
  DEV Row prep(int m) const {
    u32 f = dCin.div(m), ci = dCin.mod(m, f);
    u32 kh = dKW.div(f);
    return {xT + (i64)ci * N * Hp * Wg, int(kh), m < Mrows};
  }
  DEV int block_tap(int m0) const {   // kw of this M-tile (tile-uniform)
    u32 f = dCin.div(m0);
    u32 kh = dKW.div(f);
    return int(dKW.mod(f, kh));
  }
  DEV const u16* ptr8(const Row& rs, int k, const u16* zpage) const {
    if (!rs.valid || k >= Klim) return zpage;
    u32 n = dOHWk.div(k), r = dOHWk.mod(k, n);
    u32 oh = dWk.div(r), owp = dWk.mod(r, oh);
    return rs.cib + ((i64)n * Hp + int(oh) + rs.kh) * Wg + owp + 8;
  }
  int kcontig() const { return 0; }   // host: keep wgrad off the g8 route
};
template <> struct GldsOk<WgradXgGather> { static constexpr bool value = true; };

struct WgradDygGather {
  const u16* dyTS;         // [KW][Cout][N][OH][OWq]
  int Cout, N, OH, OWq, Klim;
  i64 copy_stride;         // Cout*N*OH*OWq elems
  MDiv dOHWk, dWk;         // SAME k-space decomposition as the A side
  struct Row { i64 off; bool valid; };
  DEV Row prep(int m) const {
    return {(i64)m * N * OH * OWq, m < Cout};
  }
  DEV const u16* ptr8_tap(const Row& rs, int k, int tap,
                          const u16* zpage) const {
    if (!rs.valid || k >= Klim) return zpage;
    u32 n = dOHWk.div(k), r = dOHWk.mod(k, n);
    u32 oh = dWk.div(r), owp = dWk.mod(r, oh);
    return dyTS + (i64)tap * copy_stride + rs.off +
           ((i64)n * OH + int(oh)) * OWq + owp + 8;
  }
  DEV const u16* ptr8(const Row& rs, int k, const u16* zpage) const {
    return ptr8_tap(rs, k, 0, zpage);
  }
};
template <> struct GldsOk<WgradDygGather> { static constexpr bool value = true; };

template <class G> struct HasTap { static constexpr bool value = false; };
template <> struct HasTap<WgradXgGather> { static constexpr bool value = true; };
template <> struct HasTap<WgradDygGather> { static constexpr bool value = true; };

// Wgrad B: virtual Bt[n'=co][k=(n,oh,ow-padded)] over dyTpad
// [Cout][N][OH][OWp] (zero-filled pad, 16B rows) — clean vector loads.
struct WgradDyGather {
  const u16* dyT;
  int Cout, N, OH, OWp, Klim;
  struct Row { const u16* rp; bool valid; };
  DEV Row prep(int m) const {
    return {dyT + (i64)m * N * OH * OWp, m < Cout};
  }
  DEV s16x8 load8(const Row& r, int k) const {
    if (r.valid && k < Klim) return *reinterpret_cast<const s16x8*>(r.rp + k);
    return zero8();
  }
};

// ------------------------------------------------------------------ core
// EPI: 0 = plain, 1 = +bias, 2 = +bias+ReLU, 3 = ReLU only.
// SPLITK: atomicAdd fp32 into c32 (epilogue applied by a finalize kernel).
// NFRAG: B fragments per wave; tile N = 32*NFRAG (4 -> 128, 2 -> 64).

// TNA/TNB: operand supplied in [reduction][col] layout (e.g. fc/1x1-conv
// wgrad, where both x and dy have the reduction as their OUTER dim); the
// transpose happens during LDS staging (scalar ds_writes) instead of a
// separate global transpose pass, and the reduction dim needs no padding.
// BM_T/THREADS_T: 128/256 (4 waves, 2x2) or 256/512 (8 waves, 4x2 — same
// per-wave 64-row tile; halves the B-panel re-read count for wgrad, whose
// dy operand is re-read once per M-tile).
// XSLAB (split-K only): grid is (slabs, tiles) instead of (tiles, slabs),
// so the hardware's round-robin XCD assignment (linear workgroup id % 8,
// x-fastest) pins every M/N-tile of one k-slab to ONE XCD when the slab
// count is a multiple of 8. Each B k-slab is then fetched from HBM once
// and re-read by the other tiles out of that XCD's L2 — the wgrad
// B-panel re-read fix (DESIGN.md "known headroom" 1).
template <class GA, class GB, int EPI, bool SPLITK, int NFRAG = 4,
          bool TNA = false, bool TNB = false, int BM_T = BM,
          int THREADS_T = THREADS, bool XSLAB = false>
__global__ __launch_bounds__(THREADS_T) void gemm_kernel(
    GA ga, GB gb, u16* __restrict__ C, float* __restrict__ c32,
    const u16* __restrict__ bias, int M, int N, int K, int k_lo, int k_hi) {
  constexpr int BN_T = 32 * NFRAG;
  __shared__ u16 As[2][BM_T * LDS_STRIDE];
  __shared__ u16 Bs[2][BN_T * LDS_STRIDE];

  const int tiles_n = (N + BN_T - 1) / BN_T;
  const int tiles_m = (M + BM_T - 1) / BM_T;
  u32 wg = XSLAB ? blockIdx.y
                 : xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int tm = wg / tiles_n, tn = wg % tiles_n;
  const int m0 = tm * BM_T, n0 = tn * BN_T;
  if (SPLITK) {
    int nchunk = XSLAB ? gridDim.x : gridDim.y;
    int slab = XSLAB ? blockIdx.x : blockIdx.y;
    int kt = (K + BK - 1) / BK;
    int span = (kt + nchunk - 1) / nchunk * BK;
    k_lo = slab * span;
    k_hi = min(K, k_lo + span);
    if (k_lo >= k_hi) return;  // empty slice: c32 is pre-zeroed
  }

  const int t = threadIdx.x;
  const int lane = t & 63, wave = t >> 6;
  const int wm = (wave >> 1) * 64, wn = (wave & 1) * (BN_T / 2);

  // staging assignment: chunk = 8 bf16; A: BM*BK/2048 chunks per thread,
  // B: BN_T*BK/2048; threads tile rows first (k-contiguous operands) or
  // k-rows first (TN operands: chunk = 8 columns of one reduction row).
  constexpr int CPT_A = BM_T * BK / 8 / THREADS_T;  // 2 at BK=32
  // B side may have fewer chunks than threads (BM_T=512/1024 threads):
  // then CPT_B=1 and only threads with brow < BN_T stage B.
  constexpr int CPT_BN = BN_T * BK / 8;             // total B chunks
  constexpr int CPT_B = CPT_BN >= THREADS_T ? CPT_BN / THREADS_T : 1;
  constexpr int TPR_A = (BK / 8) / CPT_A;           // threads per A row
  constexpr int TPR_B = (BK / 8) / CPT_B;
  const int arow = t / TPR_A, ak8 = (t % TPR_A) * (CPT_A * 8);
  const int brow = t / TPR_B, bk8 = (t % TPR_B) * (CPT_B * 8);
  const bool bstage = CPT_BN >= THREADS_T || t < CPT_BN;
  // TN assignments: chunk id -> (local k, col8)
  const int atk = (t * CPT_A) / (BM_T / 8), atc = ((t * CPT_A) % (BM_T / 8)) * 8;
  const int btk = (t * CPT_B) / (BN_T / 8), btc = ((t * CPT_B) % (BN_T / 8)) * 8;

  const auto rsa = ga.prep(TNA ? 0 : m0 + arow);
  const auto rsb = gb.prep(TNB ? 0 : n0 + brow);

  f32x4 acc[4][NFRAG];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nt = (k_hi - k_lo + BK - 1) / BK;

  s16x8 ra[CPT_A], rb[CPT_B];
  if (nt > 0) {
#pragma unroll
  for (int i = 0; i < CPT_A; ++i)
    ra[i] = TNA ? ga.load8(ga.prep(k_lo + atk), m0 + atc + i * 8)
                : ga.load8(rsa, k_lo + ak8 + i * 8);
  if (bstage)
#pragma unroll
  for (int i = 0; i < CPT_B; ++i)
    rb[i] = TNB ? gb.load8(gb.prep(k_lo + btk), n0 + btc + i * 8)
                : gb.load8(rsb, k_lo + bk8 + i * 8);
  }

  int p = 0;
  for (int tile = 0; tile < nt; ++tile) {
#pragma unroll
    for (int i = 0; i < CPT_A; ++i) {
      if (TNA) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          As[p][(atc + i * 8 + j) * LDS_STRIDE + atk] = u16(ra[i][j]);
      } else {
        *reinterpret_cast<s16x8*>(&As[p][arow * LDS_STRIDE + ak8 + i * 8]) =
            ra[i];
      }
    }
    if (bstage)
#pragma unroll
    for (int i = 0; i < CPT_B; ++i) {
      if (TNB) {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          Bs[p][(btc + i * 8 + j) * LDS_STRIDE + btk] = u16(rb[i][j]);
      } else {
        *reinterpret_cast<s16x8*>(&Bs[p][brow * LDS_STRIDE + bk8 + i * 8]) =
            rb[i];
      }
    }
    // ONE barrier per tile: also fences tile-2's readers of buf[p^1]
    __syncthreads();
    if (tile + 1 < nt) {  // issue next tile's loads under the MFMAs
      int kb = k_lo + (tile + 1) * BK;
#pragma unroll
      for (int i = 0; i < CPT_A; ++i)
        ra[i] = TNA ? ga.load8(ga.prep(kb + atk), m0 + atc + i * 8)
                    : ga.load8(rsa, kb + ak8 + i * 8);
      if (bstage)
#pragma unroll
      for (int i = 0; i < CPT_B; ++i)
        rb[i] = TNB ? gb.load8(gb.prep(kb + btk), n0 + btc + i * 8)
                    : gb.load8(rsb, kb + bk8 + i * 8);
    }
    const int kfrag = (lane >> 4) * 8;
#pragma unroll
    for (int ks = 0; ks < BK / 32; ++ks) {
      bf16x8 af[4], bf[NFRAG];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int row = wm + i * 16 + (lane & 15);
        af[i] = as_bf16x8(*reinterpret_cast<const s16x8*>(
            &As[p][row * LDS_STRIDE + ks * 32 + kfrag]));
      }
#pragma unroll
      for (int j = 0; j < NFRAG; ++j) {
        int col = wn + j * 16 + (lane & 15);
        bf[j] = as_bf16x8(*reinterpret_cast<const s16x8*>(
            &Bs[p][col * LDS_STRIDE + ks * 32 + kfrag]));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < NFRAG; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
    p ^= 1;
  }

  // ---------------------------------------------------------- epilogue
  float bvj[NFRAG];
#pragma unroll
  for (int j = 0; j < NFRAG; ++j) {
    int col = n0 + wn + j * 16 + (lane & 15);
    bvj[j] = (EPI == 1 || EPI == 2) && col < N ? bf2f(bias[col]) : 0.f;
  }
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row_base = m0 + wm + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) {
      int col = n0 + wn + j * 16 + (lane & 15);
      if (col >= N) continue;
      float bv = bvj[j];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row_base + r;
        if (row >= M) continue;
        float v = acc[i][j][r];
        if (SPLITK) {
          // fp32 atomics measure FASTER than per-slice slabs + summing
          // finalize on gfx950 (native global_atomic_add_f32; slabs cost
          // an extra write+read of sk x tile bytes) — A/B'd both ways
          atomicAdd(&c32[(i64)row * N + col], v);
        } else {
          v += bv;
          if (EPI == 2 || EPI == 3) v = v > 0.f ? v : 0.f;
          C[(i64)row * N + col] = f2bf(v);
        }
      }
    }
  }
}

// tap-aware B-source helper for the glds kernel (wgrad: block-uniform tap)
template <class GB>
DEV const u16* gb_src(const GB& gb, const typename GB::Row& r, int k,
                      int tap, const u16* zpage) {
  if constexpr (HasTap<GB>::value)
    return gb.ptr8_tap(r, k, tap, zpage);
  else
    return gb.ptr8(r, k, zpage);
}

// ------------------------------------------------- glds-staged variant
// For operands whose chunks are single 16B loads (ConvFwd/ConvDgrad/Plain):
// stage via __builtin_amdgcn_global_load_lds with PER-LANE source addresses
// (predicated-off chunks point at a 16B zero page). No staging VGPRs, no
// ds_writes; LDS is lane-linear [row][BK] (64 B rows; the residual 4-way
// ds_read_b128 conflict is the known cost of linear LDS — ladder step 3).
// ONE __shared__ arena (a second __shared__ object makes hipcc drain the
// glds queue before every ds_read — cdna_hip_programming.md section 5
// ".s-level traps" (a)); __syncthreads() itself emits the vmcnt(0) that
// completes the in-flight tile.
template <class GA, class GB, int EPI, bool SPLITK, int NFRAG = 4,
          int BKT = BK>
__global__ __launch_bounds__(THREADS) void gemm_glds_kernel(
    GA ga, GB gb, u16* __restrict__ C, float* __restrict__ c32,
    const u16* __restrict__ bias, const u16* __restrict__ zpage, int M, int N,
    int K, int k_lo, int k_hi) {
  constexpr int BN_T = 32 * NFRAG;
  constexpr int AE = BM * BKT;           // A elems per buffer
  constexpr int BE = BN_T * BKT;
  __shared__ u16 smem[2][AE + BE];

  const int tiles_n = (N + BN_T - 1) / BN_T;
  const int tiles_m = (M + BM - 1) / BM;
  u32 wg = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int tm = wg / tiles_n, tn = wg % tiles_n;
  const int m0 = tm * BM, n0 = tn * BN_T;
  if (SPLITK) {
    int nchunk = gridDim.y;
    int kt = (K + BKT - 1) / BKT;
    int span = (kt + nchunk - 1) / nchunk * BKT;
    k_lo = blockIdx.y * span;
    k_hi = min(K, k_lo + span);
    if (k_lo >= k_hi) return;
  }
  const int t = threadIdx.x;
  const int lane = t & 63, wave = t >> 6;
  const int wm = (wave >> 1) * 64, wn = (wave & 1) * (BN_T / 2);

  // per-lane glds slots: instruction g of wave w lands 1 KiB at
  // (w*GI + g)*512 elems; lane covers elems +lane*8
  constexpr int AGI = AE * 2 / 1024 / 4;   // glds instrs per wave (A)
  constexpr int BGI = BE * 2 / 1024 / 4;
  typename GA::Row rsa[AGI];
  int ak8[AGI];
  u16* alds[AGI];
#pragma unroll
  for (int g = 0; g < AGI; ++g) {
    int off = (wave * AGI + g) * 512 + lane * 8;
    rsa[g] = ga.prep(m0 + off / BKT);
    ak8[g] = off % BKT;
    alds[g] = &smem[0][(wave * AGI + g) * 512];
  }
  typename GB::Row rsb[BGI];
  int bk8[BGI];
  u16* blds[BGI];
#pragma unroll
  for (int g = 0; g < BGI; ++g) {
    int off = (wave * BGI + g) * 512 + lane * 8;
    rsb[g] = gb.prep(n0 + off / BKT);
    bk8[g] = off % BKT;
    blds[g] = &smem[0][AE + (wave * BGI + g) * 512];
  }
  constexpr int BUF = AE + BE;
  int tap = 0;
  if constexpr (HasTap<GA>::value) tap = ga.block_tap(m0);
  (void)tap;

#define STAGE_GLDS(pbuf, kb)                                                   do {                                                                           _Pragma("unroll")                                                            for (int g = 0; g < AGI; ++g)                                                  __builtin_amdgcn_global_load_lds(                                                (const __attribute__((address_space(1))) unsigned int*)                          ga.ptr8(rsa[g], (kb) + ak8[g], zpage),                                   (__attribute__((address_space(3))) unsigned int*)                                (alds[g] + (pbuf) * BUF), 16, 0, 0);                               _Pragma("unroll")                                                            for (int g = 0; g < BGI; ++g)                                                  __builtin_amdgcn_global_load_lds(                                                (const __attribute__((address_space(1))) unsigned int*)                          gb_src(gb, rsb[g], (kb) + bk8[g], tap, zpage),                                   (__attribute__((address_space(3))) unsigned int*)                                (blds[g] + (pbuf) * BUF), 16, 0, 0);                             } while (0)

  f32x4 acc[4][NFRAG];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nt = (k_hi - k_lo + BKT - 1) / BKT;
  STAGE_GLDS(0, k_lo);
  __syncthreads();          // vmcnt(0) inside: tile 0 landed

  int p = 0;
  for (int tile = 0; tile < nt; ++tile) {
    if (tile + 1 < nt) STAGE_GLDS(p ^ 1, k_lo + (tile + 1) * BKT);
    const int kfrag = (lane >> 4) * 8;
#pragma unroll
    for (int ks = 0; ks < BKT / 32; ++ks) {
      bf16x8 af[4], bf[NFRAG];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        int row = wm + i * 16 + (lane & 15);
        af[i] = as_bf16x8(*reinterpret_cast<const s16x8*>(
            &smem[p][row * BKT + ks * 32 + kfrag]));
      }
#pragma unroll
      for (int j = 0; j < NFRAG; ++j) {
        int col = wn + j * 16 + (lane & 15);
        bf[j] = as_bf16x8(*reinterpret_cast<const s16x8*>(
            &smem[p][AE + col * BKT + ks * 32 + kfrag]));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < NFRAG; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();        // drains next tile's glds; frees buf p
    p ^= 1;
  }
#undef STAGE_GLDS

  // epilogue (same as gemm_kernel)
  float bvj[NFRAG];
#pragma unroll
  for (int j = 0; j < NFRAG; ++j) {
    int col = n0 + wn + j * 16 + (lane & 15);
    bvj[j] = (EPI == 1 || EPI == 2) && col < N ? bf2f(bias[col]) : 0.f;
  }
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row_base = m0 + wm + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) {
      int col = n0 + wn + j * 16 + (lane & 15);
      if (col >= N) continue;
      float bv = bvj[j];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row_base + r;
        if (row >= M) continue;
        float v = acc[i][j][r];
        if (SPLITK) {
          atomicAdd(&c32[(i64)row * N + col], v);
        } else {
          v += bv;
          if (EPI == 2 || EPI == 3) v = v > 0.f ? v : 0.f;
          C[(i64)row * N + col] = f2bf(v);
        }
      }
    }
  }
}

// 3-buffer-ring + counted-vmcnt variant of gemm_glds_kernel (DVGG_GLDS3).
// The 2-buffer kernel drains vmcnt(0) at every tile barrier (the glds
// prefetch only spans ONE tile's MFMAs, so each tile eats most of the HBM
// latency and relies on 4-block occupancy to hide it). Here stages are
// issued TWO k-tiles ahead into a 48 KB ring (3 blocks/CU for NFRAG=4),
// the per-tile wait is a counted vmcnt(4), one raw barrier per tile
// fences the ring (write target (t+2)%3 was last read at t-1, two
// barriers back), and the LDS reads are fine-interleaved under the MFMA
// row groups with counted lgkmcnt waits (cf. gemm8 v3).
template <class GA, class GB, int EPI, bool SPLITK, int NFRAG, int BKT = 32>
__global__ __launch_bounds__(THREADS) void gemm_glds3_kernel(
    GA ga, GB gb, u16* __restrict__ C, float* __restrict__ c32,
    const u16* __restrict__ bias, const u16* __restrict__ zpage, int M, int N,
    int K, int k_lo, int k_hi) {
  constexpr int BN_T = 32 * NFRAG;
  constexpr int AE = BM * BKT;
  constexpr int BE = BN_T * BKT;
  __shared__ u16 smem[3][AE + BE];

  const int tiles_n = (N + BN_T - 1) / BN_T;
  const int tiles_m = (M + BM - 1) / BM;
  u32 wg = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int tm = wg / tiles_n, tn = wg % tiles_n;
  const int m0 = tm * BM, n0 = tn * BN_T;
  if (SPLITK) {
    int nchunk = gridDim.y;
    int kt = (K + BKT - 1) / BKT;
    int span = (kt + nchunk - 1) / nchunk * BKT;
    k_lo = blockIdx.y * span;
    k_hi = min(K, k_lo + span);
    if (k_lo >= k_hi) return;
  }
  const int t = threadIdx.x;
  const int lane = t & 63, wave = t >> 6;
  const int wm = (wave >> 1) * 64, wn = (wave & 1) * (BN_T / 2);

  constexpr int AGI = AE * 2 / 1024 / 4;
  constexpr int BGI = BE * 2 / 1024 / 4;
  typename GA::Row rsa[AGI];
  int ak8[AGI];
  int aoff[AGI];
#pragma unroll
  for (int g = 0; g < AGI; ++g) {
    int off = (wave * AGI + g) * 512 + lane * 8;
    rsa[g] = ga.prep(m0 + off / BKT);
    ak8[g] = off % BKT;
    aoff[g] = (wave * AGI + g) * 512;
  }
  typename GB::Row rsb[BGI];
  int bk8[BGI];
  int boff[BGI];
#pragma unroll
  for (int g = 0; g < BGI; ++g) {
    int off = (wave * BGI + g) * 512 + lane * 8;
    rsb[g] = gb.prep(n0 + off / BKT);
    bk8[g] = off % BKT;
    boff[g] = AE + (wave * BGI + g) * 512;
  }
  constexpr int BUF = AE + BE;
  int tap = 0;
  if constexpr (HasTap<GA>::value) tap = ga.block_tap(m0);
  (void)tap;

#define STAGE_G3(pbuf, kb)                                                     do {                                                                           _Pragma("unroll")                                                            for (int g = 0; g < AGI; ++g)                                                  __builtin_amdgcn_global_load_lds(                                                (const __attribute__((address_space(1))) unsigned int*)                          ga.ptr8(rsa[g], (kb) + ak8[g], zpage),                                   (__attribute__((address_space(3))) unsigned int*)                                (&smem[0][0] + (i64)(pbuf) * BUF + aoff[g]), 16, 0, 0);            _Pragma("unroll")                                                            for (int g = 0; g < BGI; ++g)                                                  __builtin_amdgcn_global_load_lds(                                                (const __attribute__((address_space(1))) unsigned int*)                          gb_src(gb, rsb[g], (kb) + bk8[g], tap, zpage),                                   (__attribute__((address_space(3))) unsigned int*)                                (&smem[0][0] + (i64)(pbuf) * BUF + boff[g]), 16, 0, 0);          } while (0)

  f32x4 acc[4][NFRAG];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nt = (k_hi - k_lo + BKT - 1) / BKT;
  const int kfrag = (lane >> 4) * 8;
  STAGE_G3(0, k_lo);
  if (nt > 1) STAGE_G3(1, k_lo + BKT);

  int p = 0;
  for (int tile = 0; tile < nt; ++tile) {
    // keep exactly ONE stage (the newest, AGI+BGI ops) in flight across
    // the wait — NFRAG=2 stages are 3 ops, not 4
    if (tile + 1 < nt) {
      if constexpr (AGI + BGI == 3)
        asm volatile("s_waitcnt vmcnt(3)" ::: "memory");
      else if constexpr (AGI + BGI == 4)
        asm volatile("s_waitcnt vmcnt(4)" ::: "memory");
      else
        asm volatile("s_waitcnt vmcnt(8)" ::: "memory");
    } else {
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __builtin_amdgcn_s_barrier();
    const u16* As_ = &smem[0][0] + (i64)p * BUF;
    const u16* Bs_ = As_ + AE;
#pragma unroll
    for (int ks = 0; ks < BKT / 32; ++ks) {
      bf16x8 af[4], bf[NFRAG];
#pragma unroll
      for (int j = 0; j < NFRAG; ++j) {
        int col = wn + j * 16 + (lane & 15);
        bf[j] = as_bf16x8(*reinterpret_cast<const s16x8*>(
            Bs_ + col * BKT + ks * 32 + kfrag));
      }
      af[0] = as_bf16x8(*reinterpret_cast<const s16x8*>(
          As_ + (wm + (lane & 15)) * BKT + ks * 32 + kfrag));
      asm volatile("" ::: "memory");
      if (ks == 0 && tile + 2 < nt)
        STAGE_G3((tile + 2) % 3, k_lo + (tile + 2) * BKT);
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        if (i < 3) {
          af[i + 1] = as_bf16x8(*reinterpret_cast<const s16x8*>(
              As_ + (wm + (i + 1) * 16 + (lane & 15)) * BKT + ks * 32 +
              kfrag));
          asm volatile("s_waitcnt lgkmcnt(1)" ::: "memory");
        } else {
          asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
        }
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int j = 0; j < NFRAG; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
    }
    p = p + 1 == 3 ? 0 : p + 1;
  }
#undef STAGE_G3

  // epilogue (same as gemm_glds_kernel)
  float bvj[NFRAG];
#pragma unroll
  for (int j = 0; j < NFRAG; ++j) {
    int col = n0 + wn + j * 16 + (lane & 15);
    bvj[j] = (EPI == 1 || EPI == 2) && col < N ? bf2f(bias[col]) : 0.f;
  }
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row_base = m0 + wm + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) {
      int col = n0 + wn + j * 16 + (lane & 15);
      if (col >= N) continue;
      float bv = bvj[j];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row_base + r;
        if (row >= M) continue;
        float v = acc[i][j][r];
        if (SPLITK) {
          atomicAdd(&c32[(i64)row * N + col], v);
        } else {
          v += bv;
          if (EPI == 2 || EPI == 3) v = v > 0.f ? v : 0.f;
          C[(i64)row * N + col] = f2bf(v);
        }
      }
    }
  }
}


// TN A-side gather for 1x1 STRIDED conv wgrad: reduction row m =
// (n, oh, ow) reads x at (n, oh*s, ow*s) — a strided sampling of the
// input. Used with gemm_tn_kernel (cols = ci are contiguous as in x).
struct StridedRowGather {
  const u16* x;
  int Mred, Cin, H, W, stride;
  MDiv dOHOW, dOW;
  struct Row { const u16* rp; bool valid; };
  DEV Row prep(int m) const {
    u32 n = dOHOW.div(m), r = dOHOW.mod(m, n);
    u32 oh = dOW.div(r), ow = dOW.mod(r, oh);
    return {x + (((i64)n * H + (i64)oh * stride) * W + (i64)ow * stride) *
                    Cin,
            m < Mred};
  }
  DEV s16x8 load8(const Row& r, int k) const {
    if (r.valid && k < Cin)
      return *reinterpret_cast<const s16x8*>(r.rp + k);
    return zero8();
  }
};

// ------------------------------------------------- packed-TN GEMM (K9)
// Both operands in [reduction][col] layout (fc and 1x1-conv wgrad:
// dw[M][N] = sum_k A[k][m] * B[k][n]). The generic gemm_kernel's TN mode
// transposes during staging with EIGHT scalar ds_writes per 8-elem chunk
// and measured LDS-WRITE-BOUND (41 TF on ResNet layer1 1x1 wgrads, 5x off
// the bandwidth roofline). Here each staging thread owns FOUR consecutive
// reduction rows of one 8-column group and writes the transpose as packed
// ds_write_b64 (one per column, 4x fewer write instructions); the slot
// map keeps a 16-lane write group on 8 distinct bank pairs (2-way).
// Geometry: 256 threads, 4 waves 2x2, BM=128 x BN_T=32*NFRAG, BK=32.
template <bool SPLITK, int NFRAG, class GA = PlainGather,
          class GB = PlainGather>
__global__ __launch_bounds__(THREADS) void gemm_tn_kernel(
    GA ga, GB gb, u16* __restrict__ C,
    float* __restrict__ c32, int M, int N, int K, int k_lo, int k_hi) {
  constexpr int BN_T = 32 * NFRAG;
  __shared__ u16 As[2][BM * LDS_STRIDE];
  __shared__ u16 Bs[2][BN_T * LDS_STRIDE];

  const int tiles_n = (N + BN_T - 1) / BN_T;
  const int tiles_m = (M + BM - 1) / BM;
  u32 wg = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int tm = wg / tiles_n, tn = wg % tiles_n;
  const int m0 = tm * BM, n0 = tn * BN_T;
  if (SPLITK) {
    int nchunk = gridDim.y;
    int kt = (K + BK - 1) / BK;
    int span = (kt + nchunk - 1) / nchunk * BK;
    k_lo = blockIdx.y * span;
    k_hi = min(K, k_lo + span);
    if (k_lo >= k_hi) return;
  }
  const int t = threadIdx.x;
  const int lane = t & 63, wave = t >> 6;
  const int wm = (wave >> 1) * 64, wn = (wave & 1) * (BN_T / 2);

  // staging slots: (cols/8) x (BK/4) k-quads; A on t<128, B above
  constexpr int KQ = BK / 4;               // 8
  constexpr int ASL = (BM / 8) * KQ;       // 128
  constexpr int BSL = (BN_T / 8) * KQ;     // 128 / 64
  const bool is_a = t < ASL;
  const int slot = is_a ? t : t - ASL;
  const bool active = is_a || slot < BSL;
  const int kq = slot % KQ, col8 = slot / KQ;
  const int scol = (is_a ? m0 : n0) + col8 * 8;
  const int sk0 = kq * 4;

  f32x4 acc[4][NFRAG];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nt = (k_hi - k_lo + BK - 1) / BK;
  s16x8 r[4];
  if (nt > 0 && active) {
#pragma unroll
    for (int q = 0; q < 4; ++q)
      r[q] = is_a ? ga.load8(ga.prep(k_lo + sk0 + q), scol)
                  : gb.load8(gb.prep(k_lo + sk0 + q), scol);
  }

  int p = 0;
  for (int tile = 0; tile < nt; ++tile) {
    if (active) {
      u16* base = is_a ? &As[p][(i64)(col8 * 8) * LDS_STRIDE + sk0]
                       : &Bs[p][(i64)(col8 * 8) * LDS_STRIDE + sk0];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        union { unsigned long long v; u16 e[4]; } pk;
#pragma unroll
        for (int q = 0; q < 4; ++q) pk.e[q] = u16(r[q][j]);
        *reinterpret_cast<unsigned long long*>(base + (i64)j * LDS_STRIDE) =
            pk.v;
      }
    }
    __syncthreads();
    if (tile + 1 < nt && active) {   // next tile's loads under the MFMAs
      int kb = k_lo + (tile + 1) * BK + sk0;
#pragma unroll
      for (int q = 0; q < 4; ++q)
        r[q] = is_a ? ga.load8(ga.prep(kb + q), scol)
                    : gb.load8(gb.prep(kb + q), scol);
    }
    const int kfrag = (lane >> 4) * 8;
    bf16x8 af[4], bf[NFRAG];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = wm + i * 16 + (lane & 15);
      af[i] = as_bf16x8(*reinterpret_cast<const s16x8*>(
          &As[p][row * LDS_STRIDE + kfrag]));
    }
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) {
      int col = wn + j * 16 + (lane & 15);
      bf[j] = as_bf16x8(*reinterpret_cast<const s16x8*>(
          &Bs[p][col * LDS_STRIDE + kfrag]));
    }
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < NFRAG; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[i], bf[j], acc[i][j], 0, 0, 0);
    // ONE barrier per tile: tile t+2's writes to this buffer are fenced
    // from this tile's readers by tile t+1's barrier (cf. gemm_kernel)
    p ^= 1;
  }

  // epilogue (plain or split-K atomics, EPI=0 only — wgrad has no bias)
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row_base = m0 + wm + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < NFRAG; ++j) {
      int col = n0 + wn + j * 16 + (lane & 15);
      if (col >= N) continue;
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int row = row_base + rr;
        if (row >= M) continue;
        float v = acc[i][j][rr];
        if (SPLITK) atomicAdd(&c32[(i64)row * N + col], v);
        else C[(i64)row * N + col] = f2bf(v);
      }
    }
  }
}

}  // namespace dvgg
