// 256x256-tile, 8-wave glds GEMM (the "deep" operating point of
// cdna_hip_programming.md section 5: 2-phase sync in v1; the 8-phase
// counted-vmcnt schedule is the round-2 upgrade path).
//
// Geometry: BM=BN=256, BK=64, 512 threads = 8 waves in a 2(M) x 4(N) grid;
// per-wave output 128x64 = 8x4 fragments of v_mfma_f32_16x16x32_bf16
// (64 MFMA per K-tile per wave), fp32 accumulation (128 acc VGPRs/lane).
// LDS: 2 buffers x (A 32 KB + B 32 KB) = 128 KB, one workgroup per CU.
//
// v2: counted-vmcnt k-half pipeline — each K-tile staged as two 32-KB
// k-halves issued two phases before first read; raw s_barrier + counted
// s_waitcnt vmcnt(4) keep prefetches in flight across barriers. LDS rows
// are 64 B ([row][32] per half): ds_read_b128 residual conflict <= 4-way.
#pragma once

#include "common.h"
#include "gemm_tile.h"

namespace dvgg {

constexpr int G8_BM = 256, G8_BN = 256, G8_BK = 64, G8_THREADS = 512;

// LDS elem offset for (row, kf8) in a [rows][32]-elem half with the 16 B
// segment XOR-swizzled by (row>>2)&3: a 16-lane b128 fragment read covers
// rows r..r+15 at one kf8 — unswizzled, rows r, r+4, r+8, r+12 land on the
// same bank quad (64 B rows: bank = row*16 + col dw mod 64), a structural
// 4-way conflict on EVERY fragment read (measured 37% MFMA-busy). The
// swizzle sends the 4 row-quads to the 4 distinct 16 B segments, so the 16
// lanes cover all 64 banks exactly once. glds writes stay 16 B-aligned
// (the XOR permutes whole segments) and remain conflict-free.
DEV int g8sw(int row, int kf8) {
  return row * 32 + (kf8 ^ (((row >> 2) & 3) << 3));
}

template <class GA, class GB, int EPI, bool SPLITK>
__global__ __launch_bounds__(G8_THREADS) void gemm8_kernel(
    GA ga, GB gb, u16* __restrict__ C, float* __restrict__ c32,
    const u16* __restrict__ bias, const u16* __restrict__ zpage, int M, int N,
    int K, int k_lo, int k_hi) {
  constexpr int AE = G8_BM * G8_BK;           // elems per operand buffer
  __shared__ u16 smem[2][2 * AE];             // [buf][A | B]

  const int tiles_n = (N + G8_BN - 1) / G8_BN;
  const int tiles_m = (M + G8_BM - 1) / G8_BM;
  u32 wg = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int tm = wg / tiles_n, tn = wg % tiles_n;
  const int m0 = tm * G8_BM, n0 = tn * G8_BN;
  if (SPLITK) {
    int nchunk = gridDim.y;
    int kt = (K + G8_BK - 1) / G8_BK;
    int span = (kt + nchunk - 1) / nchunk * G8_BK;
    k_lo = blockIdx.y * span;
    k_hi = min(K, k_lo + span);
    if (k_lo >= k_hi) return;
  }
  const int t = threadIdx.x;
  const int lane = t & 63, wave = t >> 6;
  const int wm = (wave >> 2) * 128, wn = (wave & 3) * 64;

  // K-half pipelined staging (v2): each K-tile's operands are staged as
  // two k-halves (A 256x32 + B 256x32 = 32 KB, 4 glds per wave each).
  // S(T,h) is issued during phase (T-1,h), i.e. TWO phases before its
  // first read, and phases wait with a COUNTED s_waitcnt vmcnt(4) + raw
  // s_barrier — prefetched halves stay in flight across barriers
  // (cdna_hip_programming.md section 5 "Pipelining across barriers";
  // __syncthreads would drain the glds queue to vmcnt(0) every phase).
  // A 16 KB k-half = 1024 sixteen-byte lane-loads = 512 threads x 2 groups
  // (the v2 bug: this loop ran g<4 as in the full-tile v1 staging, so off
  // covered 32 KB — rows 256..511 gathered garbage and aoff/boff overwrote
  // the neighbouring halves. One half-stage is 4 glds/lane, which is what
  // the vmcnt(4) phase waits below are counted for.)
  constexpr int HE = G8_BM * 32;              // elems per operand k-half
  typename GA::Row rsa[2];
  typename GB::Row rsb[2];
  int ak8[2], bk8[2];
  int aoff[2], boff[2];
#pragma unroll
  for (int g = 0; g < 2; ++g) {
    int off = (wave * 2 + g) * 1024 + lane * 16;   // byte in 16 KB half
    int row = off / 64;
    // glds LDS dests are hardware-fixed at (wave-uniform base + lane*16):
    // to realize the swizzled layout each lane FETCHES the source segment
    // that belongs at its fixed dest (seg_src = seg_dst ^ ((row>>2)&3))
    int k8 = ((off % 64) ^ (((row >> 2) & 3) << 4)) / 2;
    rsa[g] = ga.prep(m0 + row);
    ak8[g] = k8;
    aoff[g] = off / 2;                             // elem offset in A half
    rsb[g] = gb.prep(n0 + row);
    bk8[g] = k8;
    boff[g] = off / 2;
  }
  int tap = 0;
  if constexpr (HasTap<GA>::value) tap = ga.block_tap(m0);
  (void)tap;

  // LDS: [buf][khalf][A(16KB) | B(16KB)]
#define G8_HALF(pbuf, h) (&smem[0][0] + ((i64)(pbuf) * 4 + (h) * 2) * HE)
#define G8_STAGE(pbuf, h, kb)                                                  do {                                                                           _Pragma("unroll")                                                            for (int g = 0; g < 2; ++g)                                                    __builtin_amdgcn_global_load_lds(                                                (const __attribute__((address_space(1))) unsigned int*)                          ga.ptr8(rsa[g], (kb) + (h) * 32 + ak8[g], zpage),                        (__attribute__((address_space(3))) unsigned int*)                                (G8_HALF(pbuf, h) + aoff[g]), 16, 0, 0);                           _Pragma("unroll")                                                            for (int g = 0; g < 2; ++g)                                                    __builtin_amdgcn_global_load_lds(                                                (const __attribute__((address_space(1))) unsigned int*)                          gb_src(gb, rsb[g], (kb) + (h) * 32 + bk8[g], tap, zpage),                (__attribute__((address_space(3))) unsigned int*)                                (G8_HALF(pbuf, h) + HE + boff[g]), 16, 0, 0);                    } while (0)

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nt = (k_hi - k_lo + G8_BK - 1) / G8_BK;
  const int kfrag = (lane >> 4) * 8;

// v3 fine interleave: bf[0..3]+af[0..1] read up front, then each 8-MFMA
// fragment group issues the NEXT af pair before a counted lgkmcnt wait —
// LDS reads for group g+1 run under group g's MFMAs instead of bunching
// all 12 reads (and one big wait) ahead of all 32 MFMAs. The asm memory
// clobbers pin the issue order; ds reads retire in order so lgkmcnt(2)
// exactly covers "everything but the pair just issued".
#define G8_RD(dst, base, idx16) \
  dst = as_bf16x8(*reinterpret_cast<const s16x8*>((base) + (idx16)))
#define G8_PHASE(pbuf, h, WAITN, STAGE_STMT)                                   do {                                                                           asm volatile("s_waitcnt vmcnt(" #WAITN ")" ::: "memory");                    __builtin_amdgcn_s_barrier();                                                const u16* As_ = G8_HALF(pbuf, h);                                           const u16* Bs_ = As_ + HE;                                                   bf16x8 af[8], bf[4];                                                         _Pragma("unroll")                                                            for (int j = 0; j < 4; ++j)                                                    G8_RD(bf[j], Bs_, g8sw(wn + j * 16 + (lane & 15), kfrag));               G8_RD(af[0], As_, g8sw(wm + (lane & 15), kfrag));                          G8_RD(af[1], As_, g8sw(wm + 16 + (lane & 15), kfrag));                     asm volatile("" ::: "memory");                                               STAGE_STMT;                                                                  _Pragma("unroll")                                                            for (int g = 0; g < 4; ++g) {                                                  if (g < 3) {                                                                   G8_RD(af[2 * g + 2], As_,                                                        g8sw(wm + (2 * g + 2) * 16 + (lane & 15), kfrag));                   G8_RD(af[2 * g + 3], As_,                                                        g8sw(wm + (2 * g + 3) * 16 + (lane & 15), kfrag));                   asm volatile("s_waitcnt lgkmcnt(2)" ::: "memory");                         } else {                                                                     asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                         }                                                                            __builtin_amdgcn_s_setprio(1);                                               _Pragma("unroll")                                                            for (int i = 2 * g; i < 2 * g + 2; ++i)                                        _Pragma("unroll")                                                            for (int j = 0; j < 4; ++j)                                                    acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(                             af[i], bf[j], acc[i][j], 0, 0, 0);                                   __builtin_amdgcn_s_setprio(0);                                             }                                                                          } while (0)

  // prologue: T0 both halves in flight
  G8_STAGE(0, 0, k_lo);
  G8_STAGE(0, 1, k_lo);
  int p = 0;
  for (int tile = 0; tile < nt - 1; ++tile) {
    // phase (T,h): read T.h (fine-interleaved); stage (T+1).h under it
    G8_PHASE(p, 0, 4, G8_STAGE(p ^ 1, 0, k_lo + (tile + 1) * G8_BK));
    G8_PHASE(p, 1, 4, G8_STAGE(p ^ 1, 1, k_lo + (tile + 1) * G8_BK));
    p ^= 1;
  }
  // final tile: h0 waits for h1's stage still outstanding (vmcnt 4),
  // h1 drains everything (vmcnt 0)
  G8_PHASE(p, 0, 4, );
  G8_PHASE(p, 1, 0, );
#undef G8_PHASE
#undef G8_RD
#undef G8_STAGE
#undef G8_HALF
#undef G8_STAGE

  // epilogue
  float bvj[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int col = n0 + wn + j * 16 + (lane & 15);
    bvj[j] = (EPI == 1 || EPI == 2) && col < N ? bf2f(bias[col]) : 0.f;
  }
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int row_base = m0 + wm + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
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


// 256x128 (BN=128) flavor of the deep tile for N=128 layers (VGG-16
// conv2_x, ResNet layer2 3x3): 8 waves as 4(M) x 2(N), per-wave 64x64 =
// 4x4 fragments, BK=64 staged as two k-halves of (A 16 KB + B 8 KB),
// 2 ring buffers = 96 KB LDS at 1 workgroup/CU. Same counted-vmcnt
// pipeline and fine interleave as gemm8_kernel v3; one half-stage is
// 3 glds/lane (2 A groups + 1 B group), hence the vmcnt(3) waits.
template <class GA, class GB, int EPI>
__global__ __launch_bounds__(G8_THREADS) void gemm8n_kernel(
    GA ga, GB gb, u16* __restrict__ C, const u16* __restrict__ bias,
    const u16* __restrict__ zpage, int M, int N, int K) {
  constexpr int BNN = 128;
  constexpr int HEA = G8_BM * 32;             // A elems per k-half (16 KB)
  constexpr int HEB = BNN * 32;               // B elems per k-half (8 KB)
  constexpr int HB = HEA + HEB;
  __shared__ u16 smem[2][2 * HB];

  const int tiles_n = (N + BNN - 1) / BNN;
  const int tiles_m = (M + G8_BM - 1) / G8_BM;
  u32 wg = xcd_swizzle(blockIdx.x, tiles_m * tiles_n);
  const int tm = wg / tiles_n, tn = wg % tiles_n;
  const int m0 = tm * G8_BM, n0 = tn * BNN;
  const int t = threadIdx.x;
  const int lane = t & 63, wave = t >> 6;
  const int wm = (wave >> 1) * 64, wn = (wave & 1) * 64;

  typename GA::Row rsa[2];
  int ak8[2], aoff[2];
#pragma unroll
  for (int g = 0; g < 2; ++g) {
    int off = (wave * 2 + g) * 1024 + lane * 16;   // byte in 16 KB A half
    rsa[g] = ga.prep(m0 + off / 64);
    // source-side swizzle; dest stays uniform+lane*16 (see gemm8_kernel)
    ak8[g] = ((off % 64) ^ (((off / 64 >> 2) & 3) << 4)) / 2;
    aoff[g] = off / 2;
  }
  typename GB::Row rsb;
  int bk8, boff;
  {
    int off = wave * 1024 + lane * 16;             // byte in 8 KB B half
    rsb = gb.prep(n0 + off / 64);
    bk8 = ((off % 64) ^ (((off / 64 >> 2) & 3) << 4)) / 2;
    boff = off / 2;
  }
  int tap = 0;
  if constexpr (HasTap<GA>::value) tap = ga.block_tap(m0);
  (void)tap;

#define G8N_HALF(pbuf, h) (&smem[0][0] + ((i64)(pbuf) * 2 + (h)) * HB)
#define G8N_STAGE(pbuf, h, kb)                                                 do {                                                                           _Pragma("unroll")                                                            for (int g = 0; g < 2; ++g)                                                    __builtin_amdgcn_global_load_lds(                                                (const __attribute__((address_space(1))) unsigned int*)                          ga.ptr8(rsa[g], (kb) + (h) * 32 + ak8[g], zpage),                        (__attribute__((address_space(3))) unsigned int*)                                (G8N_HALF(pbuf, h) + aoff[g]), 16, 0, 0);                          __builtin_amdgcn_global_load_lds(                                                (const __attribute__((address_space(1))) unsigned int*)                          gb_src(gb, rsb, (kb) + (h) * 32 + bk8, tap, zpage),                      (__attribute__((address_space(3))) unsigned int*)                                (G8N_HALF(pbuf, h) + HEA + boff), 16, 0, 0);                     } while (0)

  f32x4 acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nt = (K + G8_BK - 1) / G8_BK;
  const int kfrag = (lane >> 4) * 8;

#define G8N_RD(dst, base, idx16)                                               dst = as_bf16x8(*reinterpret_cast<const s16x8*>((base) + (idx16)))
#define G8N_PHASE(pbuf, h, WAITN, STAGE_STMT)                                  do {                                                                           asm volatile("s_waitcnt vmcnt(" #WAITN ")" ::: "memory");                    __builtin_amdgcn_s_barrier();                                                const u16* As_ = G8N_HALF(pbuf, h);                                          const u16* Bs_ = As_ + HEA;                                                  bf16x8 af[4], bf[4];                                                         _Pragma("unroll")                                                            for (int j = 0; j < 4; ++j)                                                    G8N_RD(bf[j], Bs_, g8sw(wn + j * 16 + (lane & 15), kfrag));              G8N_RD(af[0], As_, g8sw(wm + (lane & 15), kfrag));                         asm volatile("" ::: "memory");                                               STAGE_STMT;                                                                  _Pragma("unroll")                                                            for (int g = 0; g < 4; ++g) {                                                  if (g < 3) {                                                                   G8N_RD(af[g + 1], As_,                                                           g8sw(wm + (g + 1) * 16 + (lane & 15), kfrag));                       asm volatile("s_waitcnt lgkmcnt(1)" ::: "memory");                         } else {                                                                     asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");                         }                                                                            __builtin_amdgcn_s_setprio(1);                                               _Pragma("unroll")                                                            for (int j = 0; j < 4; ++j)                                                    acc[g][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(                             af[g], bf[j], acc[g][j], 0, 0, 0);                                   __builtin_amdgcn_s_setprio(0);                                             }                                                                          } while (0)

  G8N_STAGE(0, 0, 0);
  G8N_STAGE(0, 1, 0);
  int p = 0;
  for (int tile = 0; tile < nt - 1; ++tile) {
    G8N_PHASE(p, 0, 3, G8N_STAGE(p ^ 1, 0, (tile + 1) * G8_BK));
    G8N_PHASE(p, 1, 3, G8N_STAGE(p ^ 1, 1, (tile + 1) * G8_BK));
    p ^= 1;
  }
  G8N_PHASE(p, 0, 3, );
  G8N_PHASE(p, 1, 0, );
#undef G8N_PHASE
#undef G8N_RD
#undef G8N_STAGE
#undef G8N_HALF

  // epilogue
  float bvj[4];
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int col = n0 + wn + j * 16 + (lane & 15);
    bvj[j] = (EPI == 1 || EPI == 2) && col < N ? bf2f(bias[col]) : 0.f;
  }
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int row_base = m0 + wm + i * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int col = n0 + wn + j * 16 + (lane & 15);
      if (col >= N) continue;
      float bv = bvj[j];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = row_base + r;
        if (row >= M) continue;
        float v = acc[i][j][r] + bv;
        if (EPI == 2 || EPI == 3) v = v > 0.f ? v : 0.f;
        C[(i64)row * N + col] = f2bf(v);
      }
    }
  }
}

}  // namespace dvgg
