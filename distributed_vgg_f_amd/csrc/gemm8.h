// 256x256-tile, 8-wave glds GEMM (the "deep" operating point of
// cdna_hip_programming.md section 5: 2-phase sync in v1; the 8-phase
// counted-vmcnt schedule is the round-2 upgrade path).
//
// Geometry: BM=BN=256, BK=64, 512 threads = 8 waves in a 2(M) x 4(N) grid;
// per-wave output 128x64 = 8x4 fragments of v_mfma_f32_16x16x32_bf16
// (64 MFMA per K-tile per wave), fp32 accumulation (128 acc VGPRs/lane).
// LDS: 2 buffers x (A 32 KB + B 32 KB) = 128 KB, one workgroup per CU.
//
// The LDS image uses the st_16x32 XOR swizzle (byte ^= ((byte>>9)&1)<<5):
// 128-B rows read by ds_read_b128 would otherwise be an 8-way bank
// conflict (Guideline 4). global_load_lds writes lane-linear, so the
// swizzle is applied to the per-lane SOURCE element index and repeated on
// the read side (section 5.4 rule 21; the map is an involution).
#pragma once

#include "common.h"
#include "gemm_tile.h"

namespace dvgg {

constexpr int G8_BM = 256, G8_BN = 256, G8_BK = 64, G8_THREADS = 512;

DEV int sw16x32(int byte) { return byte ^ (((byte >> 9) & 1) << 5); }

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

  // glds slots: operand bytes = 32 KB = 4 instrs/wave x 8 waves x 1 KB.
  // Linear dest byte off = (wave*4+g)*1024 + lane*16; the SOURCE element
  // comes from the swizzled position so the LDS image is st_16x32.
  typename GA::Row rsa[4];
  typename GB::Row rsb[4];
  int ak8[4], bk8[4];
  u16* alds[4];
  u16* blds[4];
#pragma unroll
  for (int g = 0; g < 4; ++g) {
    int off = (wave * 4 + g) * 1024 + lane * 16;   // dest byte (linear)
    int soff = sw16x32(off);                       // source element position
    int row = soff / 128, k8 = (soff % 128) / 2;
    rsa[g] = ga.prep(m0 + row);
    ak8[g] = k8;
    alds[g] = &smem[0][0] + off / 2;
    rsb[g] = gb.prep(n0 + row);
    bk8[g] = k8;
    blds[g] = &smem[0][0] + AE + off / 2;
  }
  int tap = 0;
  if constexpr (HasTap<GA>::value) tap = ga.block_tap(m0);
  (void)tap;

#define G8_STAGE(pbuf, kb)                                                   \
  do {                                                                       \
    _Pragma("unroll")                                                        \
    for (int g = 0; g < 4; ++g)                                              \
      __builtin_amdgcn_global_load_lds(                                      \
          (const __attribute__((address_space(1))) unsigned int*)            \
              ga.ptr8(rsa[g], (kb) + ak8[g], zpage),                         \
          (__attribute__((address_space(3))) unsigned int*)                  \
              (alds[g] + (i64)(pbuf) * 2 * AE), 16, 0, 0);                   \
    _Pragma("unroll")                                                        \
    for (int g = 0; g < 4; ++g)                                              \
      __builtin_amdgcn_global_load_lds(                                      \
          (const __attribute__((address_space(1))) unsigned int*)            \
              gb_src(gb, rsb[g], (kb) + bk8[g], tap, zpage),                 \
          (__attribute__((address_space(3))) unsigned int*)                  \
              (blds[g] + (i64)(pbuf) * 2 * AE), 16, 0, 0);                   \
  } while (0)

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int nt = (k_hi - k_lo + G8_BK - 1) / G8_BK;
  G8_STAGE(0, k_lo);
  __syncthreads();

  const int kfrag = (lane >> 4) * 8;            // 8 k-elems per lane
  int p = 0;
  for (int tile = 0; tile < nt; ++tile) {
    if (tile + 1 < nt) G8_STAGE(p ^ 1, k_lo + (tile + 1) * G8_BK);
    const u16* As = &smem[p][0];
    const u16* Bs = &smem[p][AE];
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 af[8], bf[4];
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        int row = wm + i * 16 + (lane & 15);
        int byte = sw16x32(row * 128 + (ks * 32 + kfrag) * 2);
        af[i] = as_bf16x8(
            *reinterpret_cast<const s16x8*>(As + byte / 2));
      }
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int col = wn + j * 16 + (lane & 15);
        int byte = sw16x32(col * 128 + (ks * 32 + kfrag) * 2);
        bf[j] = as_bf16x8(
            *reinterpret_cast<const s16x8*>(Bs + byte / 2));
      }
#pragma unroll
      for (int i = 0; i < 8; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
    __syncthreads();
    p ^= 1;
  }
#undef G8_STAGE

  // epilogue
#pragma unroll
  for (int i = 0; i < 8; ++i) {
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
