// Memory-bound kernels: ReLU backward (K10), bias grad (K11), residual
// add+ReLU (K18), Philox dropout (K12), fused flat-arena SGD+momentum (K14),
// fp32<->bf16 cast utilities (K16). All bf16 I/O vectorized 8-wide
// (cdna_hip_programming.md Guideline 13).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include <unordered_map>

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

// grid-stride over 8-element bf16 chunks; tail handled scalar by last chunk
#define CHUNK_LOOP(total8)                                                   \
  for (i64 c = (i64)blockIdx.x * blockDim.x + threadIdx.x; c < (total8);     \
       c += (i64)gridDim.x * blockDim.x)

static inline int grid_for(i64 chunks) {
  i64 g = (chunks + 255) / 256;
  // clamp to >=1: numel<8 would otherwise launch 0 blocks and skip the
  // scalar tail (handled by blockIdx.x==0), leaving the output garbage
  return int(std::min<i64>(std::max<i64>(g, 1), 2048));
}

// ------------------------------------------------------------- relu bwd
__global__ void relu_bwd_kernel(const u16* __restrict__ dy,
                                const u16* __restrict__ y,
                                u16* __restrict__ out, i64 n) {
  CHUNK_LOOP(n / 8) {
    s16x8 d = *reinterpret_cast<const s16x8*>(dy + c * 8);
    s16x8 v = *reinterpret_cast<const s16x8*>(y + c * 8);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      o[j] = (bf2f(u16(v[j])) > 0.f) ? d[j] : short(0);
    *reinterpret_cast<s16x8*>(out + c * 8) = o;
  }
  i64 tail = n & 7;
  if (tail && blockIdx.x == 0 && threadIdx.x < tail) {
    i64 i = n - tail + threadIdx.x;
    out[i] = (bf2f(y[i]) > 0.f) ? dy[i] : u16(0);
  }
}

at::Tensor relu_bwd(const at::Tensor& dy, const at::Tensor& y) {
  check_bf16(dy, "dy"); check_bf16(y, "y");
  auto out = at::empty_like(dy);
  i64 n = dy.numel();
  hipLaunchKernelGGL(relu_bwd_kernel, dim3(grid_for(n / 8)), dim3(256), 0,
                     cur_stream(), bfp(dy), bfp(y), bfp_mut(out), n);
  return out;
}

// ------------------------------------------------------------- add+ReLU
__global__ void add_relu_kernel(const u16* __restrict__ a,
                                const u16* __restrict__ b,
                                u16* __restrict__ out, i64 n) {
  CHUNK_LOOP(n / 8) {
    s16x8 x = *reinterpret_cast<const s16x8*>(a + c * 8);
    s16x8 y = *reinterpret_cast<const s16x8*>(b + c * 8);
    s16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = bf2f(u16(x[j])) + bf2f(u16(y[j]));
      o[j] = short(f2bf(v > 0.f ? v : 0.f));
    }
    *reinterpret_cast<s16x8*>(out + c * 8) = o;
  }
  i64 tail = n & 7;
  if (tail && blockIdx.x == 0 && threadIdx.x < tail) {
    i64 i = n - tail + threadIdx.x;
    float v = bf2f(a[i]) + bf2f(b[i]);
    out[i] = f2bf(v > 0.f ? v : 0.f);
  }
}

at::Tensor add_relu_fwd(const at::Tensor& a, const at::Tensor& b) {
  check_bf16(a, "a"); check_bf16(b, "b");
  auto out = at::empty_like(a);
  i64 n = a.numel();
  hipLaunchKernelGGL(add_relu_kernel, dim3(grid_for(n / 8)), dim3(256), 0,
                     cur_stream(), bfp(a), bfp(b), bfp_mut(out), n);
  return out;
}

// ------------------------------------------------------------- bias grad
// db[c] = sum over rows of dy [R][C]; fp32 atomics then cast.
__global__ void bias_grad_kernel(const u16* __restrict__ dy,
                                 float* __restrict__ db32, i64 R, int C,
                                 int rows_per_block) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  i64 r0 = (i64)blockIdx.y * rows_per_block;
  i64 r1 = min(R, r0 + rows_per_block);
  // 4 independent accumulators: break the serial add chain (latency-bound)
  float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
  i64 r = r0;
  for (; r + 4 <= r1; r += 4) {
    a0 += bf2f(dy[r * C + c]);
    a1 += bf2f(dy[(r + 1) * C + c]);
    a2 += bf2f(dy[(r + 2) * C + c]);
    a3 += bf2f(dy[(r + 3) * C + c]);
  }
  for (; r < r1; ++r) a0 += bf2f(dy[r * C + c]);
  atomicAdd(&db32[c], (a0 + a1) + (a2 + a3));
}

// cast + re-zero: db32 is a cached per-C scratch accumulator (atomics),
// so zeroing it here (free: the line is already in L2) replaces a
// launch-latency-bound Fill kernel before every fc bias grad.
__global__ void cast_f32_bf16_kernel(float* __restrict__ in,
                                     u16* __restrict__ out, i64 n) {
  i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) { out[i] = f2bf(in[i]); in[i] = 0.f; }
}

at::Tensor bias_grad(const at::Tensor& dy,
                     const c10::optional<at::Tensor>& db_out) {
  check_bf16(dy, "dy");
  int C = dy.size(-1);
  i64 R = dy.numel() / C;
  static std::unordered_map<int, at::Tensor> cache;  // zeroed between uses
  auto it = cache.find(C);
  if (it == cache.end())
    it = cache.emplace(C, at::zeros({C}, dy.options().dtype(at::kFloat)))
             .first;
  auto db32 = it->second;
  int rpb = 256;  // small row chunks: fill 256 CUs even at C=64
  dim3 grid(ceil_div(C, 256), ceil_div(R, rpb));
  hipLaunchKernelGGL(bias_grad_kernel, grid, dim3(256), 0, cur_stream(),
                     bfp(dy), db32.data_ptr<float>(), R, C, rpb);
  auto db = db_out ? *db_out : at::empty({C}, dy.options());
  hipLaunchKernelGGL(cast_f32_bf16_kernel, dim3(ceil_div(C, 256)), dim3(256),
                     0, cur_stream(), db32.data_ptr<float>(), bfp_mut(db), C);
  return db;
}

// ------------------------------------------------------------ SGD (K14)
// One pass over the four flat arenas:
//   g32 = scale * g (+ wd * w32); m = mu * m + g32; w32 -= lr * m;
//   wc = bf16(w32)
__global__ void sgd_kernel(float* __restrict__ w32, u16* __restrict__ wc,
                           const u16* __restrict__ g, float* __restrict__ m32,
                           i64 n, float lr, float mu, float wd, float scale) {
  // 8 elems per iteration: every access is a b128, and the two w/m pairs
  // issue back-to-back for memory-level parallelism (20 B/elem streamed;
  // the kernel is a pure HBM roofline problem)
  CHUNK_LOOP(n / 8) {
    f32x4 w0 = *reinterpret_cast<f32x4*>(w32 + c * 8);
    f32x4 w1 = *reinterpret_cast<f32x4*>(w32 + c * 8 + 4);
    f32x4 m0 = *reinterpret_cast<f32x4*>(m32 + c * 8);
    f32x4 m1 = *reinterpret_cast<f32x4*>(m32 + c * 8 + 4);
    s16x8 gv = *reinterpret_cast<const s16x8*>(g + c * 8);
    s16x8 wo;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gj = bf2f(u16(gv[j])) * scale + wd * w0[j];
      m0[j] = mu * m0[j] + gj;
      w0[j] -= lr * m0[j];
      wo[j] = short(f2bf(w0[j]));
      float gk = bf2f(u16(gv[4 + j])) * scale + wd * w1[j];
      m1[j] = mu * m1[j] + gk;
      w1[j] -= lr * m1[j];
      wo[4 + j] = short(f2bf(w1[j]));
    }
    *reinterpret_cast<f32x4*>(w32 + c * 8) = w0;
    *reinterpret_cast<f32x4*>(w32 + c * 8 + 4) = w1;
    *reinterpret_cast<f32x4*>(m32 + c * 8) = m0;
    *reinterpret_cast<f32x4*>(m32 + c * 8 + 4) = m1;
    *reinterpret_cast<s16x8*>(wc + c * 8) = wo;
  }
}

// Nontemporal variant (DVGG_SGD_NT=1): every stream is touch-once, so
// the hints bypass L2 allocation for the 1.2 GB pass.
__global__ void sgd_nt_kernel(float* __restrict__ w32, u16* __restrict__ wc,
                              const u16* __restrict__ g,
                              float* __restrict__ m32, i64 n, float lr,
                              float mu, float wd, float scale) {
  CHUNK_LOOP(n / 8) {
    f32x4 w0 = __builtin_nontemporal_load(
        reinterpret_cast<const f32x4*>(w32 + c * 8));
    f32x4 w1 = __builtin_nontemporal_load(
        reinterpret_cast<const f32x4*>(w32 + c * 8 + 4));
    f32x4 m0 = __builtin_nontemporal_load(
        reinterpret_cast<const f32x4*>(m32 + c * 8));
    f32x4 m1 = __builtin_nontemporal_load(
        reinterpret_cast<const f32x4*>(m32 + c * 8 + 4));
    s16x8 gv = __builtin_nontemporal_load(
        reinterpret_cast<const s16x8*>(g + c * 8));
    s16x8 wo;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      float gj = bf2f(u16(gv[j])) * scale + wd * w0[j];
      m0[j] = mu * m0[j] + gj;
      w0[j] -= lr * m0[j];
      wo[j] = short(f2bf(w0[j]));
      float gk = bf2f(u16(gv[4 + j])) * scale + wd * w1[j];
      m1[j] = mu * m1[j] + gk;
      w1[j] -= lr * m1[j];
      wo[4 + j] = short(f2bf(w1[j]));
    }
    __builtin_nontemporal_store(w0, reinterpret_cast<f32x4*>(w32 + c * 8));
    __builtin_nontemporal_store(w1, reinterpret_cast<f32x4*>(w32 + c * 8 + 4));
    __builtin_nontemporal_store(m0, reinterpret_cast<f32x4*>(m32 + c * 8));
    __builtin_nontemporal_store(m1, reinterpret_cast<f32x4*>(m32 + c * 8 + 4));
    __builtin_nontemporal_store(wo, reinterpret_cast<s16x8*>(wc + c * 8));
  }
}

void bump_weight_gen();  // gemm_ops.hip: weight-transform cache generation

void sgd_step(at::Tensor& w32, at::Tensor& wc, const at::Tensor& g,
              at::Tensor& m32, double lr, double mu, double wd, double scale) {
  TORCH_CHECK(w32.is_cuda() && w32.scalar_type() == at::kFloat);
  check_bf16(wc, "wc"); check_bf16(g, "g");
  i64 n = w32.numel();
  TORCH_CHECK(n % 8 == 0, "arena must be 8-aligned");  // _ALIGN=128 in arena
  static const bool nt = [] {
    const char* e = getenv("DVGG_SGD_NT");
    return e && e[0] == '1';
  }();
  if (nt)
    hipLaunchKernelGGL(sgd_nt_kernel, dim3(grid_for(n / 8)), dim3(256), 0,
                       cur_stream(), w32.data_ptr<float>(), bfp_mut(wc),
                       bfp(g), m32.data_ptr<float>(), n, float(lr), float(mu),
                       float(wd), float(scale));
  else
    hipLaunchKernelGGL(sgd_kernel, dim3(grid_for(n / 8)), dim3(256), 0,
                       cur_stream(), w32.data_ptr<float>(), bfp_mut(wc),
                       bfp(g), m32.data_ptr<float>(), n, float(lr), float(mu),
                       float(wd), float(scale));
  bump_weight_gen();  // wc changed: cached wT/wD/w8 refresh on next forward
}

// ----------------------------------------------------------- dropout K12
// Philox4x32-10 counter-based RNG: stateless, mask saved as u8.
DEV void philox_round(u32& c0, u32& c1, u32& c2, u32& c3, u32 k0, u32 k1) {
  u32 h0 = __umulhi(0xD2511F53u, c0), l0 = 0xD2511F53u * c0;
  u32 h1 = __umulhi(0xCD9E8D57u, c2), l1 = 0xCD9E8D57u * c2;
  u32 n0 = h1 ^ c1 ^ k0, n1 = l1, n2 = h0 ^ c3 ^ k1, n3 = l0;
  c0 = n0; c1 = n1; c2 = n2; c3 = n3;
}

DEV u32x4 philox(u32 lo, u32 hi, u32 key0, u32 key1) {
  u32 c0 = lo, c1 = hi, c2 = key1, c3 = 0x243F6A88u;
  u32 k0 = key0, k1 = 0xBB67AE85u;
#pragma unroll
  for (int r = 0; r < 10; ++r) {
    philox_round(c0, c1, c2, c3, k0, k1);
    k0 += 0x9E3779B9u; k1 += 0xBB67AE85u;
  }
  return u32x4{c0, c1, c2, c3};
}

__global__ void dropout_fwd_kernel(const u16* __restrict__ x,
                                   u16* __restrict__ y,
                                   unsigned char* __restrict__ mask, i64 n,
                                   float p, float inv_keep, u32 seed,
                                   u32 offset) {
  CHUNK_LOOP((n + 3) / 4) {
    u32x4 r = philox(u32(c), u32(c >> 32), seed, offset);
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      i64 i = c * 4 + j;
      if (i >= n) break;
      bool keep = (r[j] * 2.3283064365386963e-10f) >= p;
      mask[i] = keep;
      y[i] = keep ? f2bf(bf2f(x[i]) * inv_keep) : u16(0);
    }
  }
}

__global__ void dropout_bwd_kernel(const u16* __restrict__ dy,
                                   const unsigned char* __restrict__ mask,
                                   u16* __restrict__ dx, i64 n,
                                   float inv_keep) {
  i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) dx[i] = mask[i] ? f2bf(bf2f(dy[i]) * inv_keep) : u16(0);
}

std::vector<at::Tensor> dropout_fwd(const at::Tensor& x, double p,
                                    int64_t seed, int64_t offset) {
  check_bf16(x, "x");
  auto y = at::empty_like(x);
  auto mask = at::empty(x.sizes(), x.options().dtype(at::kByte));
  i64 n = x.numel();
  hipLaunchKernelGGL(dropout_fwd_kernel, dim3(grid_for((n + 3) / 4)),
                     dim3(256), 0, cur_stream(), bfp(x), bfp_mut(y),
                     mask.data_ptr<unsigned char>(), n, float(p),
                     float(1.0 / (1.0 - p)), u32(seed), u32(offset));
  return {y, mask};
}

at::Tensor dropout_bwd(const at::Tensor& dy, const at::Tensor& mask,
                       double p) {
  check_bf16(dy, "dy");
  auto dx = at::empty_like(dy);
  i64 n = dy.numel();
  hipLaunchKernelGGL(dropout_bwd_kernel, dim3(int((n + 255) / 256)),
                     dim3(256), 0, cur_stream(), bfp(dy),
                     mask.data_ptr<unsigned char>(), bfp_mut(dx), n,
                     float(1.0 / (1.0 - p)));
  return dx;
}

}  // namespace dvgg
