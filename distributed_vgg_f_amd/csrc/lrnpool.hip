// LRN and 3x3/s2 maxpool (K6/K7), NHWC.
//
// LRN design notes (MI355X): channels are innermost (NHWC), so the
// cross-channel window is a register sliding window per (n,h,w) position,
// 8-wide vectorized. A naive powf costs 2-3 transcendentals per element
// (~0.5 ms/step at bs=256 against a ~60 us bandwidth bound), so s^-beta and
// s^-(1+beta) are evaluated with a plain-VALU exp2/log2 polynomial
// (~1e-3 rel error, far below bf16 resolution) and BOTH are saved in fwd —
// backward then needs no transcendentals at all:
//   y  = x * sb,            sb  = s^-beta
//   dx = dy*sb - (2*alpha*beta/n) * x * sum_win(dy * x * snb),
//   snb = s^-(1+beta),      s = k + alpha/n * sum_win(x^2)

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace dvgg {

static inline hipStream_t cur_stream() {
  return at::hip::getCurrentHIPStreamMasqueradingAsCUDA().stream();
}
static inline const u16* bfp(const at::Tensor& t) {
  return reinterpret_cast<const u16*>(t.data_ptr<at::BFloat16>());
}
static inline u16* bfp_mut(at::Tensor& t) {
  return reinterpret_cast<u16*>(t.data_ptr<at::BFloat16>());
}
static void check_bf16(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kBFloat16 &&
              t.is_contiguous(), name, " must be contiguous GPU bf16");
}

// ---- plain-VALU log2/exp2 (no v_log/v_exp: ~4x issue rate) --------------
DEV float fast_log2(float s) {
  union { float f; u32 u; } t;
  t.f = s;
  int e = int((t.u >> 23) & 255u) - 127;
  t.u = (t.u & 0x007fffffu) | 0x3f800000u;  // m in [1,2)
  float m = t.f;
  if (m > 1.41421356f) { m *= 0.5f; e += 1; }
  float f = m - 1.0f;  // in [-0.293, 0.414)
  float p = f * (1.f - f * (0.5f - f * (0.33333333f - f * (0.25f - f * 0.2f))));
  return float(e) + p * 1.44269504f;
}

DEV float fast_exp2(float y) {
  float n = floorf(y);
  float f = y - n;
  float p = 1.f + f * (0.69314718f + f * (0.24022651f + f * (0.05550411f +
            f * (0.00961813f + f * 0.00133336f))));
  union { float f; u32 u; } t;
  t.u = u32(int(n) + 127) << 23;
  return p * t.f;
}

// ------------------------------------------------------------- LRN fwd
// one thread per (n,h,w); C % 8 == 0. Saves NOTHING: backward recomputes
// s (5-channel window sum of x^2, cheap FMAs over x it reloads anyway)
// and sb/snb via the same fast_log2/fast_exp2 polynomials, so fwd is a
// pure 2-stream (x in, y out) kernel and bwd a 3-stream one — vs the
// round-1 design that stored+reloaded sb and snb (4- and 5-stream).
__global__ void lrn_fwd_kernel(const u16* __restrict__ x, u16* __restrict__ y,
                               i64 npos, int C, float an, float k,
                               float beta) {
  // one thread per (position, 8-channel chunk): fully parallel, neighbor
  // chunks re-read through L1/L2
  i64 id = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  int nch = C / 8;
  if (id >= npos * nch) return;
  int ch = int(id % nch);
  i64 pos = id / nch;
  const u16* xr = x + pos * C + ch * 8;
  float q[12];  // squares of channels ch*8-2 .. ch*8+9
  s16x8 cur = *reinterpret_cast<const s16x8*>(xr);
  {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(u16(cur[j]));
      q[2 + j] = f * f;
    }
    q[0] = q[1] = 0.f;
    if (ch > 0) {
      float a = bf2f(xr[-2]), b = bf2f(xr[-1]);
      q[0] = a * a; q[1] = b * b;
    }
    q[10] = q[11] = 0.f;
    if (ch + 1 < nch) {
      float a = bf2f(xr[8]), b = bf2f(xr[9]);
      q[10] = a * a; q[11] = b * b;
    }
  }
  s16x8 yv;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float sum = q[j] + q[j + 1] + q[j + 2] + q[j + 3] + q[j + 4];
    float sv = k + an * sum;
    float sb = fast_exp2(-beta * fast_log2(sv));
    yv[j] = short(f2bf(bf2f(u16(cur[j])) * sb));
  }
  *reinterpret_cast<s16x8*>(y + pos * C + ch * 8) = yv;
}

// Shuffle variants: thread (pos, ch) sits next to (pos, ch+-1) in the
// wave whenever nch = C/8 divides 64, so the 2 (fwd) / 4+2 (bwd) neighbor
// channel values live in the ADJACENT LANES' registers — cross-lane
// ds_bpermute (__shfl_up/down) replaces the per-chunk scalar VMEM loads
// that made the plain kernels instruction-bound (fwd measured 2.3x off
// the 2-stream roofline; each 2 B gather is a full 64-address VMEM op).
__global__ void lrn_fwd_shfl_kernel(const u16* __restrict__ x,
                                    u16* __restrict__ y, i64 npos, int C,
                                    float an, float k, float beta) {
  i64 id = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  int nch = C / 8;
  bool live = id < npos * nch;
  id = live ? id : npos * nch - 1;        // keep the wave whole for shfl
  int ch = int(id % nch);
  i64 o = (id / nch) * C + ch * 8;
  s16x8 cur = *reinterpret_cast<const s16x8*>(x + o);
  u32 hi = u32(u16(cur[6])) | (u32(u16(cur[7])) << 16);
  u32 lo = u32(u16(cur[0])) | (u32(u16(cur[1])) << 16);
  u32 lf = __shfl_up(hi, 1);
  u32 rt = __shfl_down(lo, 1);
  float q[12];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float f = bf2f(u16(cur[j]));
    q[2 + j] = f * f;
  }
  q[0] = q[1] = 0.f;
  if (ch > 0) {
    float a = bf2f(u16(lf & 0xffff)), b = bf2f(u16(lf >> 16));
    q[0] = a * a; q[1] = b * b;
  }
  q[10] = q[11] = 0.f;
  if (ch + 1 < nch) {
    float a = bf2f(u16(rt & 0xffff)), b = bf2f(u16(rt >> 16));
    q[10] = a * a; q[11] = b * b;
  }
  s16x8 yv;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float sum = q[j] + q[j + 1] + q[j + 2] + q[j + 3] + q[j + 4];
    float sv = k + an * sum;
    float sb = fast_exp2(-beta * fast_log2(sv));
    yv[j] = short(f2bf(bf2f(u16(cur[j])) * sb));
  }
  if (live) *reinterpret_cast<s16x8*>(y + o) = yv;
}

__global__ void lrn_bwd_shfl_kernel(const u16* __restrict__ dy,
                                    const u16* __restrict__ x,
                                    u16* __restrict__ dx, i64 npos, int C,
                                    float an, float k, float beta,
                                    float coef) {
  i64 id = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  int nch = C / 8;
  bool live = id < npos * nch;
  id = live ? id : npos * nch - 1;
  int ch = int(id % nch);
  i64 o = (id / nch) * C + ch * 8;
  s16x8 dv = *reinterpret_cast<const s16x8*>(dy + o);
  s16x8 xv = *reinterpret_cast<const s16x8*>(x + o);
  // x^2 for ch*8-4..+11 and dy*x for ch*8-2..+9, edges via shuffles
  float q[16], dyx[12];
  float xf[8], df[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    xf[j] = bf2f(u16(xv[j]));
    df[j] = bf2f(u16(dv[j]));
    q[4 + j] = xf[j] * xf[j];
    dyx[2 + j] = df[j] * xf[j];
  }
  u32 xhi0 = u32(u16(xv[4])) | (u32(u16(xv[5])) << 16);
  u32 xhi1 = u32(u16(xv[6])) | (u32(u16(xv[7])) << 16);
  u32 xlo0 = u32(u16(xv[0])) | (u32(u16(xv[1])) << 16);
  u32 xlo1 = u32(u16(xv[2])) | (u32(u16(xv[3])) << 16);
  u32 lf0 = __shfl_up(xhi0, 1), lf1 = __shfl_up(xhi1, 1);
  u32 rt0 = __shfl_down(xlo0, 1), rt1 = __shfl_down(xlo1, 1);
  u32 p6 = __float_as_uint(dyx[8]), p7 = __float_as_uint(dyx[9]);
  u32 p0 = __float_as_uint(dyx[2]), p1 = __float_as_uint(dyx[3]);
  u32 lp0 = __shfl_up(p6, 1), lp1 = __shfl_up(p7, 1);
  u32 rp0 = __shfl_down(p0, 1), rp1 = __shfl_down(p1, 1);
  q[0] = q[1] = q[2] = q[3] = 0.f;
  dyx[0] = dyx[1] = 0.f;
  if (ch > 0) {
    float a = bf2f(u16(lf0 & 0xffff)), b = bf2f(u16(lf0 >> 16));
    float c = bf2f(u16(lf1 & 0xffff)), d = bf2f(u16(lf1 >> 16));
    q[0] = a * a; q[1] = b * b; q[2] = c * c; q[3] = d * d;
    dyx[0] = __uint_as_float(lp0); dyx[1] = __uint_as_float(lp1);
  }
  q[12] = q[13] = q[14] = q[15] = 0.f;
  dyx[10] = dyx[11] = 0.f;
  if (ch + 1 < nch) {
    float a = bf2f(u16(rt0 & 0xffff)), b = bf2f(u16(rt0 >> 16));
    float c = bf2f(u16(rt1 & 0xffff)), d = bf2f(u16(rt1 >> 16));
    q[12] = a * a; q[13] = b * b; q[14] = c * c; q[15] = d * d;
    dyx[10] = __uint_as_float(rp0); dyx[11] = __uint_as_float(rp1);
  }
  float t[12], sbj[8];
#pragma unroll
  for (int cc = 0; cc < 12; ++cc) {
    float sv = k + an * (q[cc] + q[cc + 1] + q[cc + 2] + q[cc + 3] +
                         q[cc + 4]);
    float sb = fast_exp2(-beta * fast_log2(sv));
    t[cc] = dyx[cc] * sb * __builtin_amdgcn_rcpf(sv);
    if (cc >= 2 && cc < 10) sbj[cc - 2] = sb;
  }
  s16x8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float wsum = t[j] + t[j + 1] + t[j + 2] + t[j + 3] + t[j + 4];
    float v = df[j] * sbj[j] - coef * xf[j] * wsum;
    out[j] = short(f2bf(v));
  }
  if (live) *reinterpret_cast<s16x8*>(dx + o) = out;
}

at::Tensor lrn_fwd(const at::Tensor& x, int64_t size, double alpha,
                   double beta, double k) {
  check_bf16(x, "x");
  TORCH_CHECK(size == 5, "LRN kernel is specialized for size=5");
  int C = x.size(-1);
  TORCH_CHECK(C % 8 == 0, "LRN needs C % 8 == 0");
  i64 npos = x.numel() / C;
  auto y = at::empty_like(x);
  i64 nthreads = npos * (C / 8);
  if (64 % (C / 8) == 0)
    hipLaunchKernelGGL(lrn_fwd_shfl_kernel, dim3(int((nthreads + 255) / 256)),
                       dim3(256), 0, cur_stream(), bfp(x), bfp_mut(y), npos,
                       C, float(alpha / size), float(k), float(beta));
  else
    hipLaunchKernelGGL(lrn_fwd_kernel, dim3(int((nthreads + 255) / 256)),
                       dim3(256), 0, cur_stream(), bfp(x), bfp_mut(y), npos,
                       C, float(alpha / size), float(k), float(beta));
  return y;
}

// ------------------------------------------------------------- LRN bwd
// dx_c = dy_c * sb_c - coef * x_c * sum_{m in win(c)} dy_m * x_m * snb_m.
// sb/snb recomputed from x: needs s for channels ch*8-2..ch*8+9, hence
// x^2 for channels ch*8-4..ch*8+11 (16 values, +-4 neighbor reads served
// by L1/L2 since adjacent chunks load those lines anyway).
__global__ void lrn_bwd_kernel(const u16* __restrict__ dy,
                               const u16* __restrict__ x,
                               u16* __restrict__ dx, i64 npos, int C,
                               float an, float k, float beta, float coef) {
  i64 id = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  int nch = C / 8;
  if (id >= npos * nch) return;
  int ch = int(id % nch);
  i64 pos = id / nch;
  i64 o = pos * C + ch * 8;
  const u16* dyr = dy + o;
  const u16* xr = x + o;
  s16x8 dv = *reinterpret_cast<const s16x8*>(dyr);
  s16x8 xv = *reinterpret_cast<const s16x8*>(xr);
  float q[16];    // x^2 for channels ch*8-4 .. ch*8+11
  float dyx[12];  // dy*x for channels ch*8-2 .. ch*8+9 (0 out of range)
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float f = bf2f(u16(xv[j]));
    q[4 + j] = f * f;
    dyx[2 + j] = bf2f(u16(dv[j])) * f;
  }
  q[0] = q[1] = q[2] = q[3] = 0.f;
  dyx[0] = dyx[1] = 0.f;
  if (ch > 0) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float f = bf2f(xr[j - 4]);
      q[j] = f * f;
    }
    dyx[0] = bf2f(dyr[-2]) * bf2f(xr[-2]);
    dyx[1] = bf2f(dyr[-1]) * bf2f(xr[-1]);
  }
  q[12] = q[13] = q[14] = q[15] = 0.f;
  dyx[10] = dyx[11] = 0.f;
  if (ch + 1 < nch) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float f = bf2f(xr[8 + j]);
      q[12 + j] = f * f;
    }
    dyx[10] = bf2f(dyr[8]) * bf2f(xr[8]);
    dyx[11] = bf2f(dyr[9]) * bf2f(xr[9]);
  }
  float t[12];   // dy*x*snb for channels ch*8-2 .. ch*8+9
  float sbj[8];  // sb for the own 8 channels
#pragma unroll
  for (int cc = 0; cc < 12; ++cc) {
    // channel ch*8-2+cc: q window q[cc..cc+4] (q index m = channel+4-ch*8)
    float sv = k + an * (q[cc] + q[cc + 1] + q[cc + 2] + q[cc + 3] +
                         q[cc + 4]);
    float sb = fast_exp2(-beta * fast_log2(sv));
    t[cc] = dyx[cc] * sb * __builtin_amdgcn_rcpf(sv);
    if (cc >= 2 && cc < 10) sbj[cc - 2] = sb;
  }
  s16x8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float wsum = t[j] + t[j + 1] + t[j + 2] + t[j + 3] + t[j + 4];
    float v = bf2f(u16(dv[j])) * sbj[j] - coef * bf2f(u16(xv[j])) * wsum;
    out[j] = short(f2bf(v));
  }
  *reinterpret_cast<s16x8*>(dx + o) = out;
}

at::Tensor lrn_bwd(const at::Tensor& dy, const at::Tensor& x, int64_t size,
                   double alpha, double beta, double k) {
  check_bf16(dy, "dy");
  int C = dy.size(-1);
  i64 npos = dy.numel() / C;
  auto dx = at::empty_like(dy);
  i64 nthreads = npos * (C / 8);
  if (64 % (C / 8) == 0)
    hipLaunchKernelGGL(lrn_bwd_shfl_kernel,
                       dim3(int((nthreads + 255) / 256)), dim3(256), 0,
                       cur_stream(), bfp(dy), bfp(x), bfp_mut(dx), npos, C,
                       float(alpha / size), float(k), float(beta),
                       float(2.0 * alpha * beta / size));
  else
    hipLaunchKernelGGL(lrn_bwd_kernel, dim3(int((nthreads + 255) / 256)),
                       dim3(256), 0, cur_stream(), bfp(dy), bfp(x),
                       bfp_mut(dx), npos, C, float(alpha / size), float(k),
                       float(beta), float(2.0 * alpha * beta / size));
  return dx;
}

// ----------------------------------------------------------- maxpool fwd
// thread per output element; saves u8 window-local argmax for backward.
__global__ void maxpool_fwd_kernel(const u16* __restrict__ x,
                                   u16* __restrict__ y,
                                   unsigned char* __restrict__ idx, int N,
                                   int H, int W, int C, int OH, int OW,
                                   int kernel, int stride, int pl) {
  // 8 channels per thread, vectorized window reads
  i64 id = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  i64 total = (i64)N * OH * OW * (C / 8);
  if (id >= total) return;
  int c8 = int(id % (C / 8)) * 8;
  i64 r = id / (C / 8);
  int ow = int(r % OW); r /= OW;
  int oh = int(r % OH); r /= OH;
  int n = int(r);
  int ih0 = oh * stride - pl, iw0 = ow * stride - pl;
  float best[8];
  int bi[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) { best[j] = -3.4e38f; bi[j] = 0; }
  for (int dh = 0; dh < kernel; ++dh) {
    int ih = ih0 + dh;
    if ((unsigned)ih >= (unsigned)H) continue;
    for (int dw = 0; dw < kernel; ++dw) {
      int iw = iw0 + dw;
      if ((unsigned)iw >= (unsigned)W) continue;
      s16x8 v = *reinterpret_cast<const s16x8*>(
          x + (((i64)n * H + ih) * W + iw) * C + c8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(u16(v[j]));
        if (f > best[j]) { best[j] = f; bi[j] = dh * kernel + dw; }
      }
    }
  }
  i64 o = (((i64)n * OH + oh) * OW + ow) * C + c8;
  s16x8 out;
  u32 i0 = 0, i1 = 0;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    out[j] = short(f2bf(best[j]));
    if (j < 4) i0 |= u32(bi[j]) << (j * 8);
    else i1 |= u32(bi[j]) << ((j - 4) * 8);
  }
  *reinterpret_cast<s16x8*>(y + o) = out;
  *reinterpret_cast<u32*>(idx + o) = i0;
  *reinterpret_cast<u32*>(idx + o + 4) = i1;
}

std::vector<at::Tensor> maxpool_fwd(const at::Tensor& x, int64_t kernel,
                                    int64_t stride, int64_t pl, int64_t pr) {
  check_bf16(x, "x");
  int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  TORCH_CHECK(H == W, "maxpool_fwd assumes square inputs (H==W), got ",
              H, "x", W);
  int OH = (H + int(pl) + int(pr) - int(kernel)) / int(stride) + 1;
  int OW = OH;  // square inputs (checked above)
  auto y = at::empty({N, OH, OW, C}, x.options());
  auto idx = at::empty({N, OH, OW, C}, x.options().dtype(at::kByte));
  TORCH_CHECK(C % 8 == 0, "maxpool needs C % 8 == 0");
  i64 total = (i64)N * OH * OW * (C / 8);
  hipLaunchKernelGGL(maxpool_fwd_kernel, dim3(int((total + 255) / 256)),
                     dim3(256), 0, cur_stream(), bfp(x), bfp_mut(y),
                     idx.data_ptr<unsigned char>(), N, H, W, C, OH, OW,
                     int(kernel), int(stride), int(pl));
  return {y, idx};
}

// k=3/s=2 specialization (every zoo pool): each input position is
// covered by at most 2x2 windows at (ohs>>1 - a, ows>>1 - b). The
// generic double loop diverges lane-by-lane (lanes of one wave span
// neighbouring positions with different window counts); here all four
// candidates run predicated with uniform control flow.
__global__ void maxpool_bwd_k3s2_kernel(const u16* __restrict__ dy,
                                        const unsigned char* __restrict__ idx,
                                        u16* __restrict__ dx, int N, int H,
                                        int W, int C, int OH, int OW,
                                        int pl) {
  i64 id = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  i64 total = (i64)N * H * W * (C / 8);
  if (id >= total) return;
  int c8 = int(id % (C / 8)) * 8;
  i64 r = id / (C / 8);
  int iw = int(r % W); r /= W;
  int ih = int(r % H); r /= H;
  int n = int(r);
  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;
  int ohs = ih + pl, ows = iw + pl;
#pragma unroll
  for (int a = 0; a < 2; ++a) {
    int oh = (ohs >> 1) - a;
    int dh = ohs - 2 * oh;
    bool okh = oh >= 0 && oh < OH && dh <= 2;
#pragma unroll
    for (int b = 0; b < 2; ++b) {
      int ow = (ows >> 1) - b;
      int dw = ows - 2 * ow;
      bool ok = okh && ow >= 0 && ow < OW && dw <= 2;
      i64 o = ok ? (((i64)n * OH + oh) * OW + ow) * C + c8 : 0;
      unsigned want = ok ? unsigned(dh * 3 + dw) : 255u;
      const unsigned char* ip = idx + o;
      u32 i0 = *reinterpret_cast<const u32*>(ip);
      u32 i1 = *reinterpret_cast<const u32*>(ip + 4);
      s16x8 d = *reinterpret_cast<const s16x8*>(dy + o);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        unsigned iv = ((j < 4 ? i0 : i1) >> ((j & 3) * 8)) & 0xffu;
        if (iv == want) acc[j] += bf2f(u16(d[j]));
      }
    }
  }
  s16x8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = short(f2bf(acc[j]));
  *reinterpret_cast<s16x8*>(dx + (((i64)n * H + ih) * W + iw) * C + c8) = out;
}

// ----------------------------------------------------------- maxpool bwd
// gather form (no atomics): each dx element scans its covering windows
// (3x3 s2 windows overlap).
__global__ void maxpool_bwd_kernel(const u16* __restrict__ dy,
                                   const unsigned char* __restrict__ idx,
                                   u16* __restrict__ dx, int N, int H, int W,
                                   int C, int OH, int OW, int kernel,
                                   int stride, int pl) {
  // 8 channels per thread (C % 8 == 0 path), vectorized loads of dy / idx
  i64 id = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  i64 total = (i64)N * H * W * (C / 8);
  if (id >= total) return;
  int c8 = int(id % (C / 8)) * 8;
  i64 r = id / (C / 8);
  int iw = int(r % W); r /= W;
  int ih = int(r % H); r /= H;
  int n = int(r);
  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;
  int oh_hi = (ih + pl) / stride;
  int ow_hi = (iw + pl) / stride;
  for (int oh = oh_hi; oh >= 0 && oh * stride > ih + pl - kernel; --oh) {
    if (oh >= OH) continue;
    int dh = ih - (oh * stride - pl);
    for (int ow = ow_hi; ow >= 0 && ow * stride > iw + pl - kernel; --ow) {
      if (ow >= OW) continue;
      int dw = iw - (ow * stride - pl);
      i64 o = (((i64)n * OH + oh) * OW + ow) * C + c8;
      unsigned char want = (unsigned char)(dh * kernel + dw);
      // idx 8 bytes + dy 16 bytes, vector loads
      const unsigned char* ip = idx + o;
      u32 i0 = *reinterpret_cast<const u32*>(ip);
      u32 i1 = *reinterpret_cast<const u32*>(ip + 4);
      s16x8 d = *reinterpret_cast<const s16x8*>(dy + o);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        unsigned char iv = (unsigned char)(((j < 4 ? i0 : i1) >> ((j & 3) * 8)) & 0xff);
        if (iv == want) acc[j] += bf2f(u16(d[j]));
      }
    }
  }
  s16x8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = short(f2bf(acc[j]));
  *reinterpret_cast<s16x8*>(dx + (((i64)n * H + ih) * W + iw) * C + c8) = out;
}

at::Tensor maxpool_bwd(const at::Tensor& dy, const at::Tensor& idx,
                       int64_t H_, int64_t W_, int64_t kernel, int64_t stride,
                       int64_t pl, int64_t pr) {
  check_bf16(dy, "dy");
  int N = dy.size(0), OH = dy.size(1), OW = dy.size(2), C = dy.size(3);
  int H = int(H_), W = int(W_);
  TORCH_CHECK(C % 8 == 0, "maxpool bwd needs C % 8 == 0");
  auto dx = at::empty({N, H, W, C}, dy.options());
  i64 total = (i64)N * H * W * (C / 8);
  if (kernel == 3 && stride == 2) {
    hipLaunchKernelGGL(maxpool_bwd_k3s2_kernel,
                       dim3(int((total + 255) / 256)), dim3(256), 0,
                       cur_stream(), bfp(dy),
                       idx.data_ptr<unsigned char>(), bfp_mut(dx), N, H, W,
                       C, OH, OW, int(pl));
    return dx;
  }
  hipLaunchKernelGGL(maxpool_bwd_kernel, dim3(int((total + 255) / 256)),
                     dim3(256), 0, cur_stream(), bfp(dy),
                     idx.data_ptr<unsigned char>(), bfp_mut(dx), N, H, W, C,
                     OH, OW, int(kernel), int(stride), int(pl));
  return dx;
}

}  // namespace dvgg
