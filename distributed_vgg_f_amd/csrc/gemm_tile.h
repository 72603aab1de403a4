// MFMA bf16 GEMM core for gfx950 — the single engine behind the FC layers
//  and all implicit-GEMM convolutions (K1/K2/K3): the
// "A"/"B" operands are gather functors, so conv fwd/dgrad/wgrad reuse one
// tile pipeline with different address functions.
//
// Geometry: 128x128 block tile, BK=32 k-step, 256 threads = 4 waves in a
// 2x2 grid, each wave computing a 64x64 sub-tile as 4x4 fragments of
// v_mfma_f32_32x... (16x16x32_bf16), fp32 accumulation in AGPRs.
// LDS tiles are [row][BK] row-major with +8 bf16 row pad (80 B row stride:
// 16 consecutive rows land on distinct bank groups for ds_read_b128 —
// cdna_hip_programming.md section 6 Guideline 4).
//
// Staging is register-staged (global->VGPR->LDS) with the next tile's
// global loads issued before the current tile's MFMAs (T14 issue-early /
// write-late, cdna_hip_programming.md Guideline 15). Layout contracts:
// every gather's k-chunks (8 consecutive virtual-k) must be either fully
// valid 16B-aligned vector loads or zero; see the functors.
#pragma once

#include "common.h"

namespace dvgg {

constexpr int BM = 128, BN = 128, BK = 32;
constexpr int LDS_PAD = 8;                 // bf16 elements
constexpr int LDS_STRIDE = BK + LDS_PAD;   // 40 elems = 80 B rows
constexpr int THREADS = 256;

DEV bf16x8 as_bf16x8(s16x8 v) {
  union { s16x8 s; bf16x8 b; } u;
  u.s = v;
  return u.b;
}

// ---------------------------------------------------------------- gathers
// Contract: load8(m, k) returns elements [m][k..k+7] of the virtual
// matrix, ZERO where out of range. k is a multiple of 8.

struct PlainGather {
  const u16* p;
  int rows, cols, ld;  // ld multiple of 8 (16B-aligned rows)
  DEV s16x8 load8(int m, int k) const {
    if (m < rows && k < cols)
      return *reinterpret_cast<const s16x8*>(p + (i64)m * ld + k);
    return s16x8{0, 0, 0, 0, 0, 0, 0, 0};
  }
};

// Implicit-GEMM forward A: virtual A[m=(n,oh,ow)][k=(kh,kw,ci)] = x NHWC.
// Requires Cin % 8 == 0 (chunks stay inside one filter tap) — conv1-style
// Cin=3 goes through im2col instead (SURVEY.md section 7 hard part 2).
struct ConvFwdGather {
  const u16* x;
  int M, H, W, Cin, OH, OW, stride, pad;
  DEV s16x8 load8(int m, int k) const {
    int f = k / Cin, c0 = k % Cin;      // Cin % 8 == 0 keeps chunk in-tap
    int kw = 0, kh = 0;
    // KW folded into f decomposition by caller packing: f = kh*KW + kw
    kh = f / KW_;
    kw = f % KW_;
    int n = m / (OH * OW), r = m % (OH * OW);
    int oh = r / OW, ow = r % OW;
    int ih = oh * stride - pad + kh, iw = ow * stride - pad + kw;
    if (m < M && (unsigned)ih < (unsigned)H && (unsigned)iw < (unsigned)W)
      return *reinterpret_cast<const s16x8*>(
          x + (((i64)n * H + ih) * W + iw) * Cin + c0);
    return s16x8{0, 0, 0, 0, 0, 0, 0, 0};
  }
  int KW_;
};

// Implicit-GEMM dgrad A: virtual A[m=(n,ih,iw)][k=(kh,kw,co)] =
// dy[n][(ih+pad-kh)/s][(iw+pad-kw)/s][co] when divisible & in range.
// Requires Cout % 8 == 0.
struct ConvDgradGather {
  const u16* dy;
  int M, H, W, Cout, OH, OW, stride, pad, KW_;
  DEV s16x8 load8(int m, int k) const {
    int f = k / Cout, c0 = k % Cout;
    int kh = f / KW_, kw = f % KW_;
    int n = m / (H * W), r = m % (H * W);
    int ih = r / W, iw = r % W;
    int ohs = ih + pad - kh, ows = iw + pad - kw;
    int oh = ohs / stride, ow = ows / stride;
    bool ok = m < M && ohs >= 0 && ows >= 0 && oh * stride == ohs &&
              ow * stride == ows && oh < OH && ow < OW;
    if (ok)
      return *reinterpret_cast<const s16x8*>(
          dy + (((i64)n * OH + oh) * OW + ow) * Cout + c0);
    return s16x8{0, 0, 0, 0, 0, 0, 0, 0};
  }
};

// Wgrad A (stride-1 convs): virtual A[m=(kh,kw,ci)][k=(n,oh,ow-padded)]
// over xTpad [Cin][N][H+2p][Wp] (halo-padded, zero-filled, built by
// nhwc2cnhw_pad). k-space uses OW rounded to OWp (mult of 8) so chunks
// never cross rows; pad reads hit pre-zeroed halo -> no predicates, but
// the +kw shift breaks 16B alignment so elements load scalar (v1; see
// csrc/README note on the wgrad staging optimization).
struct WgradXGather {
  const u16* xT;
  int Mrows, Cin, N, Hp, Wp, OH, OWp, KW_, stride;  // Hp=H+2p, Wp aligned
  DEV s16x8 load8(int m, int k) const {
    if (m >= Mrows) return s16x8{0, 0, 0, 0, 0, 0, 0, 0};
    int f = m / Cin;            // m = (kh*KW + kw) * Cin + ci
    int ci = m % Cin;
    int kh = f / KW_, kw = f % KW_;
    int n = k / (OH * OWp), r = k % (OH * OWp);
    int oh = r / OWp, ow = r % OWp;
    // stored: x[ih][iw] at [ih+pad][iw+pad]; read ih=oh*s+kh-pad -> row
    // (oh*s+kh), iw=(ow+j)*s+kw-pad -> col ((ow+j)*s+kw); halo is zero.
    const u16* base = xT + (((i64)ci * N + n) * Hp + (oh * stride + kh)) * Wp +
                      (ow * stride + kw);
    s16x8 v;
    if (stride == 1) {
      // vector staging despite the odd +kw element shift: read the aligned
      // dword window and funnel-shift (buffer has +8 elem slack for the
      // trailing read; halo is pre-zeroed so no predicates).
      uintptr_t ap = (uintptr_t)base;
      const u32* q = reinterpret_cast<const u32*>(ap & ~(uintptr_t)3);
      u32 a0 = q[0], a1 = q[1], a2 = q[2], a3 = q[3];
      u32* o = reinterpret_cast<u32*>(&v);
      if (ap & 2) {
        u32 a4 = q[4];
        o[0] = (a0 >> 16) | (a1 << 16);
        o[1] = (a1 >> 16) | (a2 << 16);
        o[2] = (a2 >> 16) | (a3 << 16);
        o[3] = (a3 >> 16) | (a4 << 16);
      } else {
        o[0] = a0; o[1] = a1; o[2] = a2; o[3] = a3;
      }
      return v;
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) v[j] = short(base[(i64)j * stride]);
    return v;
  }
};

// Forward A for Cin % 8 != 0 first layers (conv1 Cin=3, ResNet stem):
// per-element scalar gather straight from NHWC x — no im2col materialized.
// k-space is Kp = ceil8(KH*KW*Cin); elements beyond Kfull are zero.
struct ConvFwdSmallCGather {
  const u16* x;
  int M, H, W, Cin, OH, OW, stride, pad, KW_, Kfull;
  DEV s16x8 load8(int m, int k) const {
    s16x8 v{0, 0, 0, 0, 0, 0, 0, 0};
    if (m >= M) return v;
    int n = m / (OH * OW), r = m % (OH * OW);
    int oh = r / OW, ow = r % OW;
    int f = k / Cin, ci = k - f * Cin;
    int kh = f / KW_, kw = f - kh * KW_;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int kk = k + j;
      if (kk < Kfull) {
        int ih = oh * stride - pad + kh, iw = ow * stride - pad + kw;
        if ((unsigned)ih < (unsigned)H && (unsigned)iw < (unsigned)W)
          v[j] = short(x[(((i64)n * H + ih) * W + iw) * Cin + ci]);
      }
      if (++ci == Cin) {  // walk (kh,kw,ci) incrementally: no per-elem divs
        ci = 0;
        if (++kw == KW_) { kw = 0; ++kh; }
      }
    }
    return v;
  }
};

// Wgrad A for Cin % 8 != 0: scalar gather straight from NHWC x (padding via
// per-element range predicates; padded-ow k's read garbage positions but the
// B side (dyTpad) is zero there, and reads are range-clamped).
struct WgradXDirectGather {
  const u16* x;
  int Mrows, Cin, N, H, W, OH, OWp, KW_, stride, pad;
  DEV s16x8 load8(int m, int k) const {
    s16x8 v{0, 0, 0, 0, 0, 0, 0, 0};
    if (m >= Mrows) return v;
    int f = m / Cin, ci = m - f * Cin;
    int kh = f / KW_, kw = f - kh * KW_;
    int n = k / (OH * OWp), r = k % (OH * OWp);
    int oh = r / OWp, ow = r % OWp;
    int ih = oh * stride - pad + kh;
    if ((unsigned)ih >= (unsigned)H) return v;
    const u16* row = x + (((i64)n * H + ih) * W) * Cin + ci;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int iw = (ow + j) * stride - pad + kw;
      if ((unsigned)iw < (unsigned)W) v[j] = short(row[(i64)iw * Cin]);
    }
    return v;
  }
};

// Wgrad B: virtual Bt[n'=co][k=(n,oh,ow-padded)] over dyTpad
// [Cout][N][OH][OWp] (zero-filled pad, 16B rows) — clean vector loads.
struct WgradDyGather {
  const u16* dyT;
  int Cout, N, OH, OWp;
  DEV s16x8 load8(int m, int k) const {
    if (m < Cout)
      return *reinterpret_cast<const s16x8*>(dyT + (i64)m * N * OH * OWp + k);
    return s16x8{0, 0, 0, 0, 0, 0, 0, 0};
  }
};

// ------------------------------------------------------------------ core
// EPI: 0 = plain, 1 = +bias, 2 = +bias+ReLU, 3 = ReLU only.
// SPLITK: atomicAdd fp32 into c32 (epilogue applied by a finalize kernel).

template <class GA, class GB, int EPI, bool SPLITK>
__global__ __launch_bounds__(THREADS) void gemm_kernel(
    GA ga, GB gb, u16* __restrict__ C, float* __restrict__ c32,
    const u16* __restrict__ bias, int M, int N, int K, int k_lo, int k_hi) {
  __shared__ u16 As[BM * LDS_STRIDE];
  __shared__ u16 Bs[BN * LDS_STRIDE];

  const int tiles_n = (N + BN - 1) / BN;
  const int tiles_m = (M + BM - 1) / BM;
  u32 wg = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int tm = wg / tiles_n, tn = wg % tiles_n;
  const int m0 = tm * BM, n0 = tn * BN;
  if (SPLITK) {
    int nchunk = gridDim.y;
    int span = (K / BK + nchunk - 1) / nchunk * BK;
    k_lo = blockIdx.y * span;
    k_hi = min(K, k_lo + span);
    if (k_lo >= k_hi) return;
  }

  const int t = threadIdx.x;
  const int lane = t & 63, wave = t >> 6;
  const int wm = (wave >> 1) * 64, wn = (wave & 1) * 64;

  // staging assignment: thread t stages row t>>1, k-chunks (t&1)*16 + {0,8}
  const int srow = t >> 1, sk = (t & 1) * 16;

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nt = (k_hi - k_lo + BK - 1) / BK;

  // prologue: gather tile 0
  s16x8 ra0 = ga.load8(m0 + srow, k_lo + sk);
  s16x8 ra1 = ga.load8(m0 + srow, k_lo + sk + 8);
  s16x8 rb0 = gb.load8(n0 + srow, k_lo + sk);
  s16x8 rb1 = gb.load8(n0 + srow, k_lo + sk + 8);

  for (int tile = 0; tile < nt; ++tile) {
    __syncthreads();  // readers of previous tile done
    *reinterpret_cast<s16x8*>(&As[srow * LDS_STRIDE + sk]) = ra0;
    *reinterpret_cast<s16x8*>(&As[srow * LDS_STRIDE + sk + 8]) = ra1;
    *reinterpret_cast<s16x8*>(&Bs[srow * LDS_STRIDE + sk]) = rb0;
    *reinterpret_cast<s16x8*>(&Bs[srow * LDS_STRIDE + sk + 8]) = rb1;
    __syncthreads();  // tile ready
    if (tile + 1 < nt) {  // issue next tile's loads early (overlap MFMA)
      int kb = k_lo + (tile + 1) * BK;
      ra0 = ga.load8(m0 + srow, kb + sk);
      ra1 = ga.load8(m0 + srow, kb + sk + 8);
      rb0 = gb.load8(n0 + srow, kb + sk);
      rb1 = gb.load8(n0 + srow, kb + sk + 8);
    }
    // fragment reads + 16 MFMA (one K-step of 32)
    const int kfrag = (lane >> 4) * 8;
    bf16x8 af[4], bf[4];
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int row = wm + i * 16 + (lane & 15);
      af[i] = as_bf16x8(
          *reinterpret_cast<const s16x8*>(&As[row * LDS_STRIDE + kfrag]));
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = wn + j * 16 + (lane & 15);
      bf[j] = as_bf16x8(
          *reinterpret_cast<const s16x8*>(&Bs[col * LDS_STRIDE + kfrag]));
    }
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
        acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            af[i], bf[j], acc[i][j], 0, 0, 0);
  }

  // ---------------------------------------------------------- epilogue
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row_base = m0 + wm + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wn + j * 16 + (lane & 15);
      if (col >= N) continue;
      float bv = (EPI == 1 || EPI == 2) ? bf2f(bias[col]) : 0.f;
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

}  // namespace dvgg
