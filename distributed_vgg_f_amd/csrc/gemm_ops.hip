// Host launchers + auxiliary kernels for the MFMA GEMM family:
// FC layers, implicit-GEMM convolutions (K1/K2/K3),
// im2col for small-Cin first layers, transposes (K16 class), split-K
// finalize. All activations NHWC bf16, conv weights HWIO bf16.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"
#include "gemm_tile.h"
#include "gemm8.h"
#include "wgrad_fused.h"
#include "conv_ring.h"

namespace dvgg {

// defined in mem.hip
at::Tensor relu_bwd(const at::Tensor&, const at::Tensor&);
at::Tensor bias_grad(const at::Tensor&,
                     const c10::optional<at::Tensor>& = c10::nullopt);

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
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

static inline int ceil8(int v) { return (v + 7) & ~7; }

// runtime tuning knobs (benchmark sweeps; defaults are the shipped values)
static int env_int(const char* name, int dflt) {
  const char* v = getenv(name);
  return v ? atoi(v) : dflt;
}
static int nfrag_threshold() {           // N <= thresh -> 64-wide tile
  static int t = env_int("DVGG_NFRAG_THRESH", 64);
  return t;
}
static int sk_scale() {                  // percent scale on split-K targets
  static int t = env_int("DVGG_SK_SCALE", 100);
  return t;
}

// ------------------------------------------------------------ transpose2d
// [R][C] -> [C][Rp] (Rp >= R, zero-padded cols), bf16, LDS-tiled.
__global__ void transpose2d_kernel(const u16* __restrict__ in,
                                   u16* __restrict__ out, int R, int C,
                                   int Rp) {
  __shared__ u16 tile[32][33];
  int c0 = blockIdx.x * 32, r0 = blockIdx.y * 32;
  int tx = threadIdx.x & 31, ty = threadIdx.x >> 5;  // 32 x 8
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int r = r0 + ty + i * 8, c = c0 + tx;
    tile[ty + i * 8][tx] = (r < R && c < C) ? in[(i64)r * C + c] : u16(0);
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int c = c0 + ty + i * 8, r = r0 + tx;
    if (c < C && r < Rp) out[(i64)c * Rp + r] = tile[tx][ty + i * 8];
  }
}

// b128-both-sides variant (same 64x64 XOR-slot tile as nhwc2cnhw8_kernel)
// for C %% 8 == 0 && Rp %% 8 == 0 — the scalar kernel measured ~5x off
// bandwidth on the per-step fc weight transposes.
__global__ void transpose2d8_kernel(const u16* __restrict__ in,
                                    u16* __restrict__ out, int R, int C,
                                    int Rp) {
  __shared__ u16 tile8[64 * 64];
  int c0 = blockIdx.x * 64, r0 = blockIdx.y * 64;
  int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int rl = i * 32 + (t >> 3), c8 = t & 7;
    int r = r0 + rl, c = c0 + c8 * 8;
    s16x8 v = zero8();
    if (r < R && c < C)
      v = *reinterpret_cast<const s16x8*>(in + (i64)r * C + c);
    *reinterpret_cast<s16x8*>(&tile8[rl * 64 + ((c8 ^ (rl & 7)) * 8)]) = v;
  }
  __syncthreads();
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    int c = j * 32 + (t & 31), rv = (t >> 5) * 8;
    int oc = c0 + c;
    if (oc >= C || r0 + rv >= Rp) continue;
    union { s16x8 v; u16 e[8]; } pk;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      int rl = rv + k;
      pk.e[k] = tile8[rl * 64 + ((c >> 3) ^ (rl & 7)) * 8 + (c & 7)];
    }
    *reinterpret_cast<s16x8*>(out + (i64)oc * Rp + r0 + rv) = pk.v;
  }
}

static void transpose2d_into(const at::Tensor& in, at::Tensor& out) {
  int R = in.size(0), C = in.numel() / in.size(0);
  int Rp = out.size(1);
  if (C % 8 == 0 && Rp % 8 == 0) {
    dim3 grid(ceil_div(C, 64), ceil_div(Rp, 64));
    hipLaunchKernelGGL(transpose2d8_kernel, grid, dim3(256), 0, cur_stream(),
                       bfp(in), bfp_mut(out), R, C, Rp);
  } else {
    dim3 grid(ceil_div(C, 32), ceil_div(R, 32));
    hipLaunchKernelGGL(transpose2d_kernel, grid, dim3(256), 0, cur_stream(),
                       bfp(in), bfp_mut(out), R, C, Rp);
  }
}

static at::Tensor transpose2d(const at::Tensor& in, int Rp = -1) {
  int R = in.size(0), C = in.numel() / in.size(0);
  if (Rp < 0) Rp = R;
  auto out = at::zeros({C, Rp}, in.options());
  transpose2d_into(in, out);
  return out;
}

// --------------- persistent weight-transform cache (round-2 item 4)
// Weights mutate exactly once per step (sgd_step bumps the generation; the
// engine bumps via bump_weight_gen after checkpoint-load/broadcast). The
// per-forward transpose2d(w), the dgrad permute wD and the Cin<8 channel
// pad w8 then refresh once per generation instead of once per call
// (~8 transpose launches + their zero-fills per VGG-F step). Keyed by
// data_ptr and shape; OPT-IN per call (cache_w=true only for arena-homed
// weights, engine/arena.py) so ad-hoc op calls (numerics tests build fresh
// weights constantly, recycling allocator pointers) can never see a stale
// hit, and bypassed during hipGraph capture (a captured step must contain
// the refresh kernels, since replays re-run sgd_step).
i64 g_weight_gen = 0;
void bump_weight_gen() { ++g_weight_gen; }

static bool stream_capturing() {
  hipStreamCaptureStatus st = hipStreamCaptureStatusNone;
  (void)hipStreamIsCapturing(cur_stream(), &st);
  return st != hipStreamCaptureStatusNone;
}

struct WTEntry { i64 gen; std::vector<i64> src_sizes; at::Tensor t; };

// kind 0: wT [Cout][K]; kind 1: wD [Cin][KhKw*Cout]; kind 2: w8 pad
template <class F>
static at::Tensor cached_weight_form(int kind, bool on, const at::Tensor& w,
                                     F&& make) {
  if (!on || stream_capturing()) return make(c10::nullopt);
  static std::unordered_map<const void*, WTEntry> cache[5];
  auto& c = cache[kind];
  std::vector<i64> sizes(w.sizes().begin(), w.sizes().end());
  auto it = c.find(w.data_ptr());
  if (it != c.end() && it->second.src_sizes == sizes) {
    if (it->second.gen == g_weight_gen) return it->second.t;
    it->second.t = make(c10::optional<at::Tensor>(it->second.t));
    it->second.gen = g_weight_gen;
    return it->second.t;
  }
  auto t = make(c10::nullopt);
  c[w.data_ptr()] = WTEntry{g_weight_gen, std::move(sizes), t};
  return t;
}

// --------------------------------------------- NHWC -> C,N,Hp,Wp (padded)
// For the wgrad gathers: writes x[n][ih][iw][c] to out[c][n][ih+ro][iw+co];
// out pre-zeroed (halo + alignment pad).
__global__ void nhwc2cnhw_kernel(const u16* __restrict__ in,
                                 u16* __restrict__ out, int N, int H, int W,
                                 int C, int Hp, int Wp, int ro, int co) {
  __shared__ u16 tile[32][33];
  int nh = blockIdx.x;            // n*H + ih
  int w0 = blockIdx.y * 32, c0 = blockIdx.z * 32;
  int n = nh / H, ih = nh % H;
  int tx = threadIdx.x & 31, ty = threadIdx.x >> 5;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int iw = w0 + ty + i * 8, c = c0 + tx;
    tile[ty + i * 8][tx] = (iw < W && c < C)
        ? in[(((i64)n * H + ih) * W + iw) * C + c] : u16(0);
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int c = c0 + ty + i * 8, iw = w0 + tx;
    if (c < C && iw < W)
      out[(((i64)c * N + n) * Hp + (ih + ro)) * Wp + iw + co] =
          tile[tx][ty + i * 8];
  }
}

// Vectorized (b128 both sides) NHWC -> C,N,Hp,Wp transpose for C % 8 == 0.
// 64(out-col) x 64(c) tile per block, 8x fewer global memory instructions
// than the u16 path (which measured ~8x off bandwidth speed-of-light on
// VGG-16: 223 us vs ~26 us for a 103 MB dy). The tile is indexed by OUT
// column wp (so 16B stores are always aligned regardless of the halo
// offset co); the source read is shifted by -co, and out-of-image
// positions store explicit zeros into the (pre-zeroed) halo columns.
// LDS slot XOR swizzle keeps both phases <= 2-way conflicted.
__global__ void nhwc2cnhw8_kernel(const u16* __restrict__ in,
                                  u16* __restrict__ out, int N, int H, int W,
                                  int C, int Hp, int Wp, int ro, int co) {
  __shared__ u16 tile8[64 * 64];
  int nh = blockIdx.x;
  int wb = blockIdx.y * 64, c0 = blockIdx.z * 64;
  int n = nh / H, ih = nh % H;
  int t = threadIdx.x;
  const u16* inb = in + ((i64)n * H + ih) * W * C;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int wloc = i * 32 + (t >> 3), c8 = t & 7;
    int iw = wb + wloc - co, c = c0 + c8 * 8;
    s16x8 v = zero8();
    if ((unsigned)iw < (unsigned)W && c < C)
      v = *reinterpret_cast<const s16x8*>(inb + (i64)iw * C + c);
    *reinterpret_cast<s16x8*>(&tile8[wloc * 64 + ((c8 ^ (wloc & 7)) * 8)]) = v;
  }
  __syncthreads();
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    int c = j * 32 + (t & 31), wv = (t >> 5) * 8;
    int oc = c0 + c;
    if (oc >= C || wb + wv >= Wp) continue;
    union { s16x8 v; u16 e[8]; } pk;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      int w = wv + k;
      pk.e[k] = tile8[w * 64 + ((c >> 3) ^ (w & 7)) * 8 + (c & 7)];
    }
    *reinterpret_cast<s16x8*>(
        out + (((i64)oc * N + n) * Hp + (ih + ro)) * Wp + wb + wv) = pk.v;
  }
}

// Phase-plane transpose for strided small-Cin wgrad (VGG-F conv1):
// x [N,H,W,C] -> xP [C][s][s][N][Hq][Wqp] with ih = s*ihq + ph,
// iw = s*iwq + pw (s = stride, pad = 0). One block per (n, ih): stage the
// W*C row in LDS, then emit C*s phase rows of Wq elems with b128 stores.
// Turns the stride-s wgrad A gather into the stride-1 funnel gather
// (WgradXGather phl mode) — the scalar XDirect path measured ~87 TF on
// conv1, this path removes the stride-s scatter entirely.
__global__ void nhwc2phase_kernel(const u16* __restrict__ in,
                                  u16* __restrict__ out, int N, int H, int W,
                                  int C, int s, int Wq, int Hq, int Wqp,
                                  int pad) {
  extern __shared__ u16 rowbuf[];          // W*C elems (host-checked <= 4096)
  int nh = blockIdx.x;
  int n = nh / H, ih = nh % H;
  int t = threadIdx.x;
  int elems = W * C;
  const u16* src = in + (i64)nh * elems;
  for (int e8 = t * 8; e8 < elems; e8 += blockDim.x * 8) {
    if (e8 + 8 <= elems)
      *reinterpret_cast<s16x8*>(&rowbuf[e8]) =
          *reinterpret_cast<const s16x8*>(src + e8);
    else
      for (int j = e8; j < elems; ++j) rowbuf[j] = src[j];
  }
  __syncthreads();
  int ph = (ih + pad) % s, ihq = (ih + pad) / s;   // halo folded in
  int rows = C * s;                        // (c, pw) phase rows
  int chunks = (Wq + 7) / 8;
  for (int job = t; job < rows * chunks; job += blockDim.x) {
    int r = job / chunks, q = job % chunks;
    int c = r / s, pw = r % s;
    union { s16x8 v; u16 e[8]; } pk;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int iwq = q * 8 + j;
      int iw = iwq * s + pw - pad;
      pk.e[j] = (iwq < Wq && iw >= 0 && iw < W) ? rowbuf[iw * C + c]
                                                : u16(0);
    }
    i64 plane = ((i64)c * s + ph) * s + pw;
    *reinterpret_cast<s16x8*>(
        out + ((plane * N + n) * Hq + ihq) * Wqp + q * 8) = pk.v;
  }
}

// dyTS builder for the glds wgrad: KW pre-shifted transposed copies of the
// (masked) dy. Copy kw holds dy[n][oh][ow][co] at
// dyTS[kw][co][n][oh][ow + kw - pad + 8]; everything else stays zero.
__global__ void dyts_kernel(const u16* __restrict__ dy, u16* __restrict__ out,
                            int N, int OH, int OW, int C, int OWq, int KW,
                            int pad, i64 copy_stride) {
  __shared__ u16 tile[32][33];
  int nh = blockIdx.x;
  int w0 = blockIdx.y * 32, c0 = blockIdx.z * 32;
  int n = nh / OH, oh = nh % OH;
  int tx = threadIdx.x & 31, ty = threadIdx.x >> 5;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int iw = w0 + ty + i * 8, c = c0 + tx;
    tile[ty + i * 8][tx] = (iw < OW && c < C)
        ? dy[(((i64)n * OH + oh) * OW + iw) * C + c] : u16(0);
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int c = c0 + ty + i * 8, iw = w0 + tx;
    if (c < C && iw < OW) {
      u16 v = tile[tx][ty + i * 8];
      i64 base = (((i64)c * N + n) * OH + oh) * OWq + iw + 8 - pad;
      for (int kw = 0; kw < KW; ++kw)
        out[kw * copy_stride + base + kw] = v;
    }
  }
}

// Cached, zeroed, slack-padded buffers for the wgrad transposes: the
// interior is fully rewritten every call and the zero halo is never
// written, so the (expensive at bs=256) zero-fill happens once per shape.
static at::Tensor cached_zero_buffer(const std::string& key, i64 numel,
                                     const at::TensorOptions& opt) {
  static std::unordered_map<std::string, at::Tensor> cache;
  auto it = cache.find(key);
  if (it == cache.end()) {
    it = cache.emplace(key, at::zeros({numel + 8}, opt)).first;
  }
  return it->second;
}

// Scratch fp32 split-K accumulators (atomic epilogue): cached per
// (site, shape); the finalize kernel re-zeros the buffer after reading
// it, so the per-call at::zeros Fill launch disappears (~10 small Fill
// launches/step on VGG-F, each mostly launch latency).
static at::Tensor scratch_c32(const char* site, i64 R, i64 C,
                              const at::TensorOptions& opt) {
  char key[64];
  snprintf(key, sizeof key, "c32:%s:%lldx%lld", site, (long long)R,
           (long long)C);
  return cached_zero_buffer(key, R * C, opt.dtype(at::kFloat));
}

static at::Tensor nhwc2cnhw_pad(const at::Tensor& x, int Hp, int Wp, int ro,
                                int co, char role) {
  int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  i64 numel = (i64)C * N * Hp * Wp;
  char key[96];
  snprintf(key, sizeof key, "%c%d_%d_%d_%d_%d_%d_%d", role, x.get_device(),
           C, N, Hp, Wp, ro, co);
  auto out = cached_zero_buffer(key, numel, x.options())
      .narrow(0, 0, numel).view({C, N, Hp, Wp});
  if (C % 8 == 0 && Wp % 8 == 0) {
    dim3 grid(N * H, ceil_div(Wp, 64), ceil_div(C, 64));
    hipLaunchKernelGGL(nhwc2cnhw8_kernel, grid, dim3(256), 0, cur_stream(),
                       bfp(x), bfp_mut(out), N, H, W, C, Hp, Wp, ro, co);
  } else {
    dim3 grid(N * H, ceil_div(W, 32), ceil_div(C, 32));
    hipLaunchKernelGGL(nhwc2cnhw_kernel, grid, dim3(256), 0, cur_stream(),
                       bfp(x), bfp_mut(out), N, H, W, C, Hp, Wp, ro, co);
  }
  return out;
}

// Host wrapper for nhwc2phase_kernel: cached pre-zeroed planes buffer.
static at::Tensor nhwc2phase(const at::Tensor& x, int s, int Hq, int Wqp,
                             int pad) {
  int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  int Wq = ceil_div(W + 2 * pad, s);
  TORCH_CHECK(W * C <= 4096, "phase transpose row too large");
  i64 numel = (i64)C * s * s * N * Hq * Wqp;
  char key[96];
  snprintf(key, sizeof key, "q%d_%d_%d_%d_%d_%d_%d", x.get_device(), C, s, N,
           Hq, Wqp, pad);
  auto out = cached_zero_buffer(key, numel, x.options())
      .narrow(0, 0, numel).view({C * s * s, N, Hq, Wqp});
  dim3 grid(N * H);
  hipLaunchKernelGGL(nhwc2phase_kernel, grid, dim3(256),
                     (W * C + 8) * sizeof(u16), cur_stream(), bfp(x),
                     bfp_mut(out), N, H, W, C, s, Wq, Hq, Wqp, pad);
  return out;
}

__global__ void cast_f32_bf16_2(const float* __restrict__ in,
                                u16* __restrict__ out, i64 n) {
  i64 i = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) out[i] = f2bf(in[i]);
}

// db[c] = bf16( sum over 256 slots of db32p[s][c] )
// (a 1024-thread sub-stripe fold was tried and measured WORSE,
// 17.2 -> 20.9 us avg — the one-block form is already L2-served)
__global__ void bias_slot_reduce_kernel(float* __restrict__ db32p,
                                        u16* __restrict__ db, int C) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float a = 0.f;
#pragma unroll 8
  for (int s = 0; s < 256; ++s) a += db32p[(i64)s * C + c];
  db[c] = f2bf(a);
#pragma unroll 8
  for (int s = 0; s < 256; ++s) db32p[(i64)s * C + c] = 0.f;  // scratch_c32
}

// Vectorized preamble (see bwd_preamble_kernel below for semantics):
// b128 dy/y loads, b128 dym stores, 64x64 transpose tile with the same
// XOR-slot LDS layout as nhwc2cnhw8_kernel, per-thread fp32 db partials
// reduced through LDS to one atomic per channel per block. Requires
// C % 8 == 0 and Wp % 8 == 0 (always true: C is a conv Cout).
template <bool RELU, bool WRITE_DYM, bool DB>
__global__ void bwd_preamble8_kernel(const u16* __restrict__ dy,
                                     const u16* __restrict__ y,
                                     u16* __restrict__ dym,
                                     u16* __restrict__ dyT,
                                     float* __restrict__ db32, int N, int H,
                                     int W, int C, int Wp) {
  __shared__ u16 tile8[64 * 64];
  __shared__ float dbp[32][64];
  int nh = blockIdx.x;
  int wb = blockIdx.y * 64, c0 = blockIdx.z * 64;
  int n = nh / H, ih = nh % H;
  int t = threadIdx.x;
  const i64 rbase = ((i64)n * H + ih) * W * C;
  float dba[8];
  if (DB)
#pragma unroll
    for (int k = 0; k < 8; ++k) dba[k] = 0.f;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int wloc = i * 32 + (t >> 3), c8 = t & 7;
    int iw = wb + wloc, c = c0 + c8 * 8;
    union { s16x8 v; u16 e[8]; } pk;
    pk.v = zero8();
    if (iw < W && c < C) {
      i64 o = rbase + (i64)iw * C + c;
      pk.v = *reinterpret_cast<const s16x8*>(dy + o);
      if (RELU) {
        union { s16x8 v; u16 e[8]; } yv;
        yv.v = *reinterpret_cast<const s16x8*>(y + o);
#pragma unroll
        for (int k = 0; k < 8; ++k)
          if (!(bf2f(yv.e[k]) > 0.f)) pk.e[k] = 0;
      }
      if (WRITE_DYM) *reinterpret_cast<s16x8*>(dym + o) = pk.v;
      if (DB)
#pragma unroll
        for (int k = 0; k < 8; ++k) dba[k] += bf2f(pk.e[k]);
    }
    *reinterpret_cast<s16x8*>(&tile8[wloc * 64 + ((c8 ^ (wloc & 7)) * 8)]) =
        pk.v;
  }
  if (DB) {
#pragma unroll
    for (int k = 0; k < 8; ++k) dbp[t >> 3][(t & 7) * 8 + k] = dba[k];
  }
  __syncthreads();
  if (DB && t < 64 && c0 + t < C) {
    float a = 0.f;
#pragma unroll
    for (int r = 0; r < 32; ++r) a += dbp[r][t];
    atomicAdd(&db32[(i64)(blockIdx.x & 255) * C + c0 + t], a);
  }
#pragma unroll
  for (int j = 0; j < 2; ++j) {
    int c = j * 32 + (t & 31), wv = (t >> 5) * 8;
    int oc = c0 + c;
    if (oc >= C || wb + wv >= Wp) continue;
    union { s16x8 v; u16 e[8]; } pk;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      int w = wv + k;
      pk.e[k] = tile8[w * 64 + ((c >> 3) ^ (w & 7)) * 8 + (c & 7)];
    }
    *reinterpret_cast<s16x8*>(
        dyT + (((i64)oc * N + n) * H + ih) * Wp + wb + wv) = pk.v;
  }
}

// ------------------------- fused conv-backward preamble (K10+K11+K16)
// One read of dy produces: ReLU-masked dy (NHWC, for dgrad), the padded
// transpose dyTpad [C][N][OH][OWp] (for wgrad), and fp32 bias-grad partials.
template <bool RELU, bool WRITE_DYM, bool DB>
__global__ void bwd_preamble_kernel(const u16* __restrict__ dy,
                                    const u16* __restrict__ y,
                                    u16* __restrict__ dym,
                                    u16* __restrict__ dyT,
                                    float* __restrict__ db32, int N, int H,
                                    int W, int C, int Wp) {
  __shared__ u16 tile[32][33];
  __shared__ float dbred[8][32];
  int nh = blockIdx.x;
  int w0 = blockIdx.y * 32, c0 = blockIdx.z * 32;
  int n = nh / H, ih = nh % H;
  int tx = threadIdx.x & 31, ty = threadIdx.x >> 5;
  float dbacc = 0.f;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int iw = w0 + ty + i * 8, c = c0 + tx;
    u16 v = 0;
    if (iw < W && c < C) {
      i64 o = (((i64)n * H + ih) * W + iw) * C + c;
      v = dy[o];
      if (RELU && !(bf2f(y[o]) > 0.f)) v = 0;
      if (WRITE_DYM) dym[o] = v;
      if (DB) dbacc += bf2f(v);
    }
    tile[ty + i * 8][tx] = v;
  }
  if (DB) dbred[ty][tx] = dbacc;
  __syncthreads();
  // ONE atomic per channel per block, into 256-way slotted partials —
  // per-thread atomics onto C floats serialized the whole kernel
  if (DB && ty == 0 && c0 + tx < C) {
    float a = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j) a += dbred[j][tx];
    atomicAdd(&db32[(blockIdx.x & 255) * C + c0 + tx], a);
  }
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int c = c0 + ty + i * 8, iw = w0 + tx;
    if (c < C && iw < W)
      dyT[(((i64)c * N + n) * H + ih) * Wp + iw] = tile[tx][ty + i * 8];
  }
}

// ------------------------------------------------------- split-K finalize
// out[id] = cast(sum over slices of c32[z][id] (+bias[c]) (relu));
// c32 is [slices][R][C], 8 fp32 per thread per slice.
template <int EPI>
__global__ void finalize_kernel(float* __restrict__ c32,
                                u16* __restrict__ out,
                                const u16* __restrict__ bias, i64 total,
                                int C, int slices, i64 slice_stride,
                                int rezero) {
  i64 id8 = ((i64)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  if (id8 >= total) return;
  float acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) acc[j] = 0.f;
  for (int z = 0; z < slices; ++z) {
    const float* p = c32 + (i64)z * slice_stride + id8;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      if (id8 + j < total) acc[j] += p[j];
  }
  if (rezero)  // scratch_c32 contract: leave the accumulator zeroed
    for (int z = 0; z < slices; ++z) {
      float* q = c32 + (i64)z * slice_stride + id8;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (id8 + j < total) q[j] = 0.f;
    }
  for (int j = 0; j < 8; ++j) {
    i64 id = id8 + j;
    if (id >= total) break;
    float v = acc[j];
    if (EPI == 1 || EPI == 2) v += bf2f(bias[id % C]);
    if (EPI == 2 || EPI == 3) v = v > 0.f ? v : 0.f;
    out[id] = f2bf(v);
  }
}

// 1-elem/thread finalize for SMALL outputs x MANY slices (the 8-wide
// kernel would launch total/8/256 blocks — 18 blocks for a 37 MB
// 256-slice slab read on VGG-16 conv1_2): full-width parallelism, each
// wave's lanes read consecutive elems of each slice (coalesced).
__global__ void finalize_narrow_kernel(const float* __restrict__ c32,
                                       u16* __restrict__ out, i64 total,
                                       int slices, i64 slice_stride) {
  i64 id = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  if (id >= total) return;
  float a0 = 0.f, a1 = 0.f, a2 = 0.f, a3 = 0.f;
  const float* p = c32 + id;
  int z = 0;
  for (; z + 4 <= slices; z += 4) {
    a0 += p[(i64)z * slice_stride];
    a1 += p[(i64)(z + 1) * slice_stride];
    a2 += p[(i64)(z + 2) * slice_stride];
    a3 += p[(i64)(z + 3) * slice_stride];
  }
  for (; z < slices; ++z) a0 += p[(i64)z * slice_stride];
  out[id] = f2bf((a0 + a1) + (a2 + a3));
}

static void run_finalize(const at::Tensor& c32, at::Tensor& out,
                         const at::Tensor* bias, int epi, int R, int C,
                         int slices = 1, bool rezero = false) {
  i64 total = (i64)R * C;
  i64 sstride = (i64)R * C;
  int blocks = int((total / 8 + 256) / 256);
  if (epi == 0 && slices > 8 && blocks < 512) {
    hipLaunchKernelGGL(finalize_narrow_kernel,
                       dim3(int((total + 255) / 256)), dim3(256), 0,
                       cur_stream(), c32.data_ptr<float>(), bfp_mut(out),
                       total, slices, sstride);
    return;
  }
  const u16* bp = bias ? bfp(*bias) : nullptr;
  auto s = cur_stream();
  switch (epi) {
    case 0: hipLaunchKernelGGL(finalize_kernel<0>, dim3(blocks), dim3(256), 0,
                               s, c32.data_ptr<float>(), bfp_mut(out), bp,
                               total, C, slices, sstride, int(rezero)); break;
    case 1: hipLaunchKernelGGL(finalize_kernel<1>, dim3(blocks), dim3(256), 0,
                               s, c32.data_ptr<float>(), bfp_mut(out), bp,
                               total, C, slices, sstride, int(rezero)); break;
    case 2: hipLaunchKernelGGL(finalize_kernel<2>, dim3(blocks), dim3(256), 0,
                               s, c32.data_ptr<float>(), bfp_mut(out), bp,
                               total, C, slices, sstride, int(rezero)); break;
    default: hipLaunchKernelGGL(finalize_kernel<3>, dim3(blocks), dim3(256),
                                0, s, c32.data_ptr<float>(), bfp_mut(out), bp,
                                total, C, slices, sstride, int(rezero)); break;
  }
}

// ------------------------------------------------------------ gemm driver
static const u16* zero_page(const at::TensorOptions& opt) {
  return bfp(cached_zero_buffer("zpage", 64, opt));
}

template <class GA, class GB, int NFRAG>
static void launch_gemm_t(const GA& ga, const GB& gb, at::Tensor* outB,
                          at::Tensor* c32, const at::Tensor* bias, int epi,
                          int M, int N, int K, int splitk) {
  constexpr int BN_T = 32 * NFRAG;
  int tiles = ceil_div(M, BM) * ceil_div(N, BN_T);
  auto s = cur_stream();
  const u16* bp = bias ? bfp(*bias) : nullptr;
  constexpr bool GLDS = GldsOk<GA>::value && GldsOk<GB>::value;
  if constexpr (GLDS) {
    const u16* zp = zero_page(outB ? outB->options() : c32->options()
                                  .dtype(at::kBFloat16));
    u16* op = outB ? bfp_mut(*outB) : nullptr;
    float* cp = c32 ? c32->data_ptr<float>() : nullptr;
    dim3 gridk(tiles, splitk);
    dim3 grid1(tiles);
    // A/B verdict: BK=64 (64 KB LDS, 2 blocks/CU) loses 2-7% on every
    // model incl. the K=64 1x1 single-tile case — occupancy beats barrier
    // count again. Template retained for future tuning.
    bool bk64 = false;
    // 3-buffer counted-vmcnt ring variant (A/B; see gemm_glds3_kernel)
    bool g3 = env_int("DVGG_GLDS3", 1);  // measured: VGG-F -1.5%, others flat+
    // 256x256 8-wave deep tile for wide-N shapes (DVGG_GEMM8 to A/B).
    // Measured routing gates (VGG-F/VGG-16/ResNet A/B, profiles/):
    //  * fill: at 1 workgroup/CU the deep tile needs enough blocks to
    //    cover the chip; with the v3 fine interleave the break-even
    //    dropped to ~96 tiles (VGG-F conv3-5 at 169 tiles now WIN:
    //    5.04 -> 4.89 ms) (DVGG_G8_MINTILES);
    //  * contiguity: scattered A gathers (Cin=64 -> one tap per BK tile,
    //    VGG-F conv2) lose to the 4-blocks/CU 128^2 kernel, which hides
    //    gather latency with occupancy; dense K runs >= 128 win ~10-15%
    //    per call (VGG-16 fwd/dgrad) (DVGG_G8_MINCONTIG).
    // (split-K stays on the 128^2 kernel: gemm8's per-element fp32-atomic
    // split-K epilogue measured -5% on VGG-F's fc layers.)
    // BN=128 deep-tile flavor for N in [128, 256) (DVGG_GEMM8N; off by
    // default pending a perf A/B — numerics-validated): VGG-16 conv2_x
    // and ResNet layer2 shapes.
    if (NFRAG == 4 && N >= 128 && N < 256 && M >= 256 && K >= 128 &&
        splitk == 1 && env_int("DVGG_GEMM8N", 0) &&
        ga.kcontig() >= env_int("DVGG_G8_MINCONTIG", 128) &&
        ceil_div(M, G8_BM) * ceil_div(N, 128) >=
            env_int("DVGG_G8_MINTILES", 96)) {
      int tiles8n = ceil_div(M, G8_BM) * ceil_div(N, 128);
      dim3 g8(tiles8n);
      switch (epi) {
        case 0:
          hipLaunchKernelGGL((gemm8n_kernel<GA, GB, 0>), g8,
                             dim3(G8_THREADS), 0, s, ga, gb, op, nullptr,
                             zp, M, N, K);
          return;
        case 1:
          hipLaunchKernelGGL((gemm8n_kernel<GA, GB, 1>), g8,
                             dim3(G8_THREADS), 0, s, ga, gb, op, bp, zp, M,
                             N, K);
          return;
        case 2:
          hipLaunchKernelGGL((gemm8n_kernel<GA, GB, 2>), g8,
                             dim3(G8_THREADS), 0, s, ga, gb, op, bp, zp, M,
                             N, K);
          return;
        default:
          hipLaunchKernelGGL((gemm8n_kernel<GA, GB, 3>), g8,
                             dim3(G8_THREADS), 0, s, ga, gb, op, nullptr,
                             zp, M, N, K);
          return;
      }
    }
    int tiles8 = ceil_div(M, G8_BM) * ceil_div(N, G8_BN);
    if (NFRAG == 4 && N >= 256 && M >= 256 && K >= 128 && splitk == 1 &&
        env_int("DVGG_GEMM8", 1) &&
        ga.kcontig() >= env_int("DVGG_G8_MINCONTIG", 128) &&
        tiles8 >= env_int("DVGG_G8_MINTILES", 96)) {
      if (splitk > 1) {
        dim3 g8(tiles8, splitk);
        hipLaunchKernelGGL((gemm8_kernel<GA, GB, 0, true>), g8,
                           dim3(G8_THREADS), 0, s, ga, gb, nullptr, cp,
                           nullptr, zp, M, N, K, 0, K);
        return;
      }
      dim3 g8(tiles8);
      switch (epi) {
        case 0:
          hipLaunchKernelGGL((gemm8_kernel<GA, GB, 0, false>), g8,
                             dim3(G8_THREADS), 0, s, ga, gb, op, nullptr,
                             nullptr, zp, M, N, K, 0, K);
          return;
        case 1:
          hipLaunchKernelGGL((gemm8_kernel<GA, GB, 1, false>), g8,
                             dim3(G8_THREADS), 0, s, ga, gb, op, nullptr, bp,
                             zp, M, N, K, 0, K);
          return;
        case 2:
          hipLaunchKernelGGL((gemm8_kernel<GA, GB, 2, false>), g8,
                             dim3(G8_THREADS), 0, s, ga, gb, op, nullptr, bp,
                             zp, M, N, K, 0, K);
          return;
        default:
          hipLaunchKernelGGL((gemm8_kernel<GA, GB, 3, false>), g8,
                             dim3(G8_THREADS), 0, s, ga, gb, op, nullptr,
                             nullptr, zp, M, N, K, 0, K);
          return;
      }
    }
#define GLDS_DISPATCH(EPIC, SPLITC, GRID, CPTR, BPTR)                            do {                                                                           if (g3)                                                                        hipLaunchKernelGGL((gemm_glds3_kernel<GA, GB, EPIC, SPLITC, NFRAG,                                                32>), GRID, dim3(THREADS), 0,                              s, ga, gb, op, CPTR, BPTR, zp, M, N, K, 0, K);          else if (bk64)                                                                 hipLaunchKernelGGL((gemm_glds_kernel<GA, GB, EPIC, SPLITC, NFRAG,                                                 64>), GRID, dim3(THREADS), 0,                              s, ga, gb, op, CPTR, BPTR, zp, M, N, K, 0, K);          else                                                                           hipLaunchKernelGGL((gemm_glds_kernel<GA, GB, EPIC, SPLITC, NFRAG,                                                 32>), GRID, dim3(THREADS), 0,                              s, ga, gb, op, CPTR, BPTR, zp, M, N, K, 0, K);        } while (0)
    if (splitk > 1) {
      GLDS_DISPATCH(0, true, gridk, cp, nullptr);
      return;
    }
    switch (epi) {
      case 0: GLDS_DISPATCH(0, false, grid1, nullptr, nullptr); return;
      case 1: GLDS_DISPATCH(1, false, grid1, nullptr, bp); return;
      case 2: GLDS_DISPATCH(2, false, grid1, nullptr, bp); return;
      default: GLDS_DISPATCH(3, false, grid1, nullptr, nullptr); return;
    }
#undef GLDS_DISPATCH
  } else {   // register-staged path (never instantiated for glds-only gathers)
  if (splitk > 1) {
    dim3 grid(tiles, splitk);
    hipLaunchKernelGGL((gemm_kernel<GA, GB, 0, true, NFRAG>), grid,
                       dim3(THREADS), 0, s, ga, gb, nullptr,
                       c32->data_ptr<float>(), nullptr, M, N, K, 0, K);
  } else {
    dim3 grid(tiles);
    switch (epi) {
      case 0:
        hipLaunchKernelGGL((gemm_kernel<GA, GB, 0, false, NFRAG>), grid,
                           dim3(THREADS), 0, s, ga, gb, bfp_mut(*outB),
                           nullptr, nullptr, M, N, K, 0, K);
        break;
      case 1:
        hipLaunchKernelGGL((gemm_kernel<GA, GB, 1, false, NFRAG>), grid,
                           dim3(THREADS), 0, s, ga, gb, bfp_mut(*outB),
                           nullptr, bp, M, N, K, 0, K);
        break;
      case 2:
        hipLaunchKernelGGL((gemm_kernel<GA, GB, 2, false, NFRAG>), grid,
                           dim3(THREADS), 0, s, ga, gb, bfp_mut(*outB),
                           nullptr, bp, M, N, K, 0, K);
        break;
      default:
        hipLaunchKernelGGL((gemm_kernel<GA, GB, 3, false, NFRAG>), grid,
                           dim3(THREADS), 0, s, ga, gb, bfp_mut(*outB),
                           nullptr, nullptr, M, N, K, 0, K);
        break;
    }
  }
  }
}

template <class GA, class GB>
static void launch_gemm(const GA& ga, const GB& gb, at::Tensor* outB,
                        at::Tensor* c32, const at::Tensor* bias, int epi,
                        int M, int N, int K, int splitk) {
  if (N <= nfrag_threshold())
    launch_gemm_t<GA, GB, 2>(ga, gb, outB, c32, bias, epi, M, N, K, splitk);
  else
    launch_gemm_t<GA, GB, 4>(ga, gb, outB, c32, bias, epi, M, N, K, splitk);
}

// TN Plain x Plain launcher (fc / 1x1-conv wgrad): operands in
// [reduction][col] layout, transposed during LDS staging.
template <int NFRAG>
static void launch_gemm_tn_t(const PlainGather& ga, const PlainGather& gb,
                             at::Tensor* outB, at::Tensor* c32, int M, int N,
                             int K, int splitk) {
  constexpr int BN_T = 32 * NFRAG;
  int tiles = ceil_div(M, BM) * ceil_div(N, BN_T);
  auto s = cur_stream();
  if (splitk > 1) {
    dim3 grid(tiles, splitk);
    hipLaunchKernelGGL((gemm_kernel<PlainGather, PlainGather, 0, true, NFRAG,
                                    true, true>), grid, dim3(THREADS), 0, s,
                       ga, gb, nullptr, c32->data_ptr<float>(), nullptr, M, N,
                       K, 0, K);
  } else {
    hipLaunchKernelGGL((gemm_kernel<PlainGather, PlainGather, 0, false, NFRAG,
                                    true, true>), dim3(tiles), dim3(THREADS),
                       0, s, ga, gb, bfp_mut(*outB), nullptr, nullptr, M, N,
                       K, 0, K);
  }
}

template <class GA, class GB>
static void launch_gemm_tn(const GA& ga, const GB& gb,
                           at::Tensor* outB, at::Tensor* c32, int M, int N,
                           int K, int splitk) {
  constexpr bool PLAIN =
      std::is_same<GA, PlainGather>::value &&
      std::is_same<GB, PlainGather>::value;
  if (!PLAIN || env_int("DVGG_TNPACK", 1)) {
    // packed-b64 transpose staging (gemm_tn_kernel): 4x fewer LDS write
    // instructions than the scalar TN mode of gemm_kernel
    auto s = cur_stream();
    int tiles4 = ceil_div(M, BM) * ceil_div(N, BN);
    int tiles2 = ceil_div(M, BM) * ceil_div(N, 64);
    if (splitk > 1) {
      if (N <= nfrag_threshold())
        hipLaunchKernelGGL((gemm_tn_kernel<true, 2, GA, GB>),
                           dim3(tiles2, splitk), dim3(THREADS), 0, s, ga, gb,
                           nullptr, c32->data_ptr<float>(), M, N, K, 0, K);
      else
        hipLaunchKernelGGL((gemm_tn_kernel<true, 4, GA, GB>),
                           dim3(tiles4, splitk), dim3(THREADS), 0, s, ga, gb,
                           nullptr, c32->data_ptr<float>(), M, N, K, 0, K);
    } else {
      if (N <= nfrag_threshold())
        hipLaunchKernelGGL((gemm_tn_kernel<false, 2, GA, GB>), dim3(tiles2),
                           dim3(THREADS), 0, s, ga, gb, bfp_mut(*outB),
                           nullptr, M, N, K, 0, K);
      else
        hipLaunchKernelGGL((gemm_tn_kernel<false, 4, GA, GB>), dim3(tiles4),
                           dim3(THREADS), 0, s, ga, gb, bfp_mut(*outB),
                           nullptr, M, N, K, 0, K);
    }
    return;
  }
  if constexpr (PLAIN) {
    if (N <= nfrag_threshold())
      launch_gemm_tn_t<2>(ga, gb, outB, c32, M, N, K, splitk);
    else
      launch_gemm_tn_t<4>(ga, gb, outB, c32, M, N, K, splitk);
  }
}

// wgrad-dedicated launcher: BM=256/512-thread tile when Cout >= 128
// (halves the dy B-panel re-reads — wgrad's binding constraint).
template <class GA, class GB>
static void launch_gemm_wgrad(const GA& ga, const GB& gb, at::Tensor* c32,
                              int M, int N, int K, int splitk) {
  auto s = cur_stream();
  if (N >= 128 && M >= 512 && env_int("DVGG_WGRAD_BM512", 0)) {
    // A/B verdict: WORSE on every model (VGG-F 5.27->5.95, VGG-16
    // 33.0->36.9, ResNet 45.9->47.4) — 1 block/CU (100 KB LDS) loses more
    // latency hiding than the quartered dy B-panel re-reads buy, the same
    // pattern as the BK=64 and XSLAB experiments. Kept for reference.
    // 512-row tile, 16 waves: quarters the dy B-panel re-read count
    // (tiles_m) at the cost of 1 block/CU (100 KB LDS).
    int tiles = ceil_div(M, 512) * ceil_div(N, 128);
    dim3 grid(tiles, splitk);
    hipLaunchKernelGGL((gemm_kernel<GA, GB, 0, true, 4, false, false, 512,
                                    1024>), grid, dim3(1024), 0, s, ga, gb,
                       nullptr, c32->data_ptr<float>(), nullptr, M, N, K, 0,
                       K);
    return;
  }
  if (N >= 128 && M >= 256 && env_int("DVGG_WGRAD_BM256", 1)) {
    int tiles = ceil_div(M, 256) * ceil_div(N, 128);
    // A/B verdict: XSLAB measured WORSE (VGG-F 5.48->5.59, VGG-16
    // 32.6->34.5) — either the linear-id%8 XCD mapping assumption fails
    // here or slab skew/atomic finalize costs dominate; default off,
    // revisit with PMC L2 counters in round 2.
    if (env_int("DVGG_WGRAD_XSLAB", 0)) {
      // slab-major grid + slab count a multiple of 8 => every tile of one
      // k-slab runs on one XCD; the dy B-slab is HBM-fetched once and
      // re-read out of that XCD's L2 (see gemm_kernel XSLAB note).
      int sk8 = std::max(8, (splitk + 7) / 8 * 8);
      dim3 grid(sk8, tiles);
      hipLaunchKernelGGL((gemm_kernel<GA, GB, 0, true, 4, false, false, 256,
                                      512, true>), grid, dim3(512), 0, s, ga,
                         gb, nullptr, c32->data_ptr<float>(), nullptr, M, N,
                         K, 0, K);
      return;
    }
    dim3 grid(tiles, splitk);
    hipLaunchKernelGGL((gemm_kernel<GA, GB, 0, true, 4, false, false, 256,
                                    512>), grid, dim3(512), 0, s, ga, gb,
                       nullptr, c32->data_ptr<float>(), nullptr, M, N, K, 0,
                       K);
    return;
  }
  launch_gemm(ga, gb, nullptr, c32, nullptr, 0, M, N, K, splitk);
}

static int pick_splitk(int tiles, int K, int target_blocks = 512) {
  target_blocks = target_blocks * sk_scale() / 100;
  if (tiles >= target_blocks) return 1;
  int kt = std::max(1, K / BK);
  int want = std::min(kt, std::max(1, target_blocks / std::max(1, tiles)));
  // collapse to the number of NON-EMPTY slices at this span
  int span = (kt + want - 1) / want;
  return (kt + span - 1) / span;
}

static int epi_of(bool has_bias, bool relu) {
  return has_bias ? (relu ? 2 : 1) : (relu ? 3 : 0);
}

// ---------------------------------------------------------------- linear
at::Tensor linear_fwd(const at::Tensor& x, const at::Tensor& w,
                      const c10::optional<at::Tensor>& bias, bool relu,
                      bool cache_w) {
  check_bf16(x, "x"); check_bf16(w, "w");
  int M = x.size(0), K = x.size(1), N = w.size(1);
  TORCH_CHECK(w.size(0) == K && K % 8 == 0 && N % 8 == 0, "bad fc shape");
  auto y = at::empty({M, N}, x.options());
  auto wT = cached_weight_form(0, cache_w, w, [&](c10::optional<at::Tensor> prev) {
    if (prev) { transpose2d_into(w, *prev); return *prev; }
    return transpose2d(w);
  });  // [N][K]
  PlainGather ga{bfp(x), M, K, K};
  PlainGather gb{bfp(wT), N, K, K};
  int tiles = ceil_div(M, BM) * ceil_div(N, BN);
  int epi = epi_of(bias.has_value(), relu);
  int sk = pick_splitk(tiles, K);
  if (sk > 1) {
    auto c32 = scratch_c32("fcf", M, N, x.options());
    launch_gemm(ga, gb, nullptr, &c32, nullptr, 0, M, N, K, sk);
    run_finalize(c32, y, bias ? &*bias : nullptr, epi, M, N, 1, true);
  } else {
    launch_gemm(ga, gb, &y, nullptr, bias ? &*bias : nullptr, epi, M, N, K, 1);
  }
  return y;
}

at::Tensor linear_dgrad(const at::Tensor& dy, const at::Tensor& w) {
  check_bf16(dy, "dy"); check_bf16(w, "w");
  int M = dy.size(0), N = dy.size(1), K = w.size(0);
  TORCH_CHECK(w.size(1) == N && N % 8 == 0, "bad fc dgrad shape");
  auto dx = at::empty({M, K}, dy.options());
  PlainGather ga{bfp(dy), M, N, N};
  PlainGather gb{bfp(w), K, N, N};  // Bt[kin][n] = w — no transpose needed
  int tiles = ceil_div(M, BM) * ceil_div(K, BN);
  int sk = pick_splitk(tiles, N);
  if (sk > 1) {
    auto c32 = scratch_c32("fcd", M, K, dy.options());
    launch_gemm(ga, gb, nullptr, &c32, nullptr, 0, M, K, N, sk);
    run_finalize(c32, dx, nullptr, 0, M, K, 1, true);
  } else {
    launch_gemm(ga, gb, &dx, nullptr, nullptr, 0, M, K, N, 1);
  }
  return dx;
}

at::Tensor linear_wgrad(const at::Tensor& x, const at::Tensor& dy,
                        const c10::optional<at::Tensor>& dw_out) {
  check_bf16(x, "x"); check_bf16(dy, "dy");
  int M = x.size(0), K = x.size(1), N = dy.size(1);
  TORCH_CHECK(dy.size(0) == M, "bad fc wgrad shape");
  // TN-staged GEMM: x [M][K] and dy [M][N] consumed directly (reduction =
  // their outer dim, transposed in LDS during staging) — no global
  // transpose pass, no reduction padding.
  // dw_out: write straight into the caller's grad-arena view (skips the
  // autograd AccumulateGrad add; see engine/arena.py direct-grad contract).
  auto dw = dw_out ? dw_out->view({K, N}) : at::empty({K, N}, x.options());
  int Mred = ((M + BK - 1) / BK) * BK;   // k-loop granularity (rows pred'd)
  PlainGather ga{bfp(x), M, K, K};       // rows = reduction
  PlainGather gb{bfp(dy), M, N, N};
  int tiles = ceil_div(K, BM) * ceil_div(N, BN);
  int sk = pick_splitk(tiles, Mred);
  if (sk > 1) {
    auto c32 = scratch_c32("fcw", K, N, x.options());
    launch_gemm_tn(ga, gb, nullptr, &c32, K, N, Mred, sk);
    run_finalize(c32, dw, nullptr, 0, K, N, 1, true);
  } else {
    launch_gemm_tn(ga, gb, &dw, nullptr, K, N, Mred, 1);
  }
  return dw;
}

// Channel pad for Cin < 8 first layers: x [N,H,W,C] -> [N,H,W,8] with zero
// fill, so conv1/stem take the vector-gather + glds GEMM paths instead of
// scalar gathers (weights padded likewise; padded channels produce zero
// grads/outputs).
__global__ void pad_channels_kernel(const u16* __restrict__ x,
                                    u16* __restrict__ out, i64 npos, int C) {
  i64 pos = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  if (pos >= npos) return;
  const u16* src = x + pos * C;
  s16x8 v{0, 0, 0, 0, 0, 0, 0, 0};
  for (int c = 0; c < C; ++c) v[c] = short(src[c]);
  *reinterpret_cast<s16x8*>(out + pos * 8) = v;
}

static at::Tensor pad_channels8(const at::Tensor& x) {
  int N = x.size(0), H = x.size(1), W = x.size(2), C = x.size(3);
  auto out = at::empty({N, H, W, 8}, x.options());
  i64 npos = (i64)N * H * W;
  hipLaunchKernelGGL(pad_channels_kernel, dim3(int((npos + 255) / 256)),
                     dim3(256), 0, cur_stream(), bfp(x), bfp_mut(out), npos,
                     C);
  return out;
}

// ---------------- tap-fused 3x3 / 5x5 wgrad (wgrad_fused.h) -----------
static bool wgrad_fused_ok(int KH, int KW, int stride, int Cin, int Cout) {
  if (stride != 1 || Cin % 64 != 0 || Cout % 64 != 0) return false;
  if (KH == 3 && KW == 3) return env_int("DVGG_WGRAD_FUSED", 1);
  if (KH == 5 && KW == 5) return env_int("DVGG_WGRAD_FUSED5", 1);
  return false;
}

// OW -> the fused kernel's padded column extent: multiple of 8 with
// OWp2 % 32 == 0 or 32 % OWp2 == 0 (so a 32-k tile is whole rows or a
// whole 32-column segment). 56 -> 64, 112 -> 128 (14% zero-k waste).
static int owp2_of(int OW) {
  int p = ceil8(OW);
  if (p % 32 == 0 || 32 % p == 0) return p;
  return (p + 31) & ~31;
}

static void wgrad_fused(const at::Tensor& xT, const at::Tensor& dyT,
                        at::Tensor& dw2d, int KH, int KW, int Cin, int Cout,
                        int N, int Hp, int Wp, int OH, int OWp2) {
  int CT = std::min(OWp2, 32), R = 32 / CT;
  int SEG = CT == 32 ? OWp2 / 32 : 1;
  int OHK = (OH + R - 1) / R * R;
  int TPI = CT == 32 ? OH * SEG : OHK / R;
  i64 tiles_k = (i64)N * TPI;
  bool k5 = KH == 5;
  int wavesn = (!k5 && Cout % 128 == 0) ? 4 : 2;
  int tiles = (Cin / 64) * (Cout / (32 * wavesn));
  int target = env_int("DVGG_WGF_BLOCKS", 256) * sk_scale() / 100;
  // rolling-window ring variant: slices are whole (n, c0) column runs
  int LT = CT == 32 ? OH : OHK / R;        // tiles per run
  int runs_total = N * SEG;
  bool ring = env_int("DVGG_WGF_RING", 1) && LT >= 3;
  i64 sk_units = ring ? runs_total : tiles_k;
  int sk = std::max(1, std::min<int>(int(sk_units),
                                     target / std::max(1, tiles)));
  // slab budget: every slice writes a full |dw| fp32 slab (read back by
  // the summing finalize) — cap the total slab footprint
  i64 outs = (i64)KH * KW * Cin * Cout;
  i64 emax = (i64)env_int("DVGG_WGF_SLAB_MB", 256) * (1 << 20) / 4;
  sk = std::max(std::min(sk, 8), std::min<int>(sk, int(emax / outs)));
  i64 span = (sk_units + sk - 1) / sk;
  sk = int((sk_units + span - 1) / span);  // non-empty slices only
  // per-slice slabs, fully written by construction: no zero-fill
  auto c32 = at::empty({sk, (i64)KH * KW * Cin, Cout},
                       xT.options().dtype(at::kFloat));
  const u16* zp = zero_page(xT.options());
  dim3 grid(tiles, sk);
  auto s = cur_stream();
  if (ring) {
#define WGR_LAUNCH(CTV, WMV, WNV, KHV, KWV)                                  \
    hipLaunchKernelGGL((wgrad_ring_kernel<CTV, WMV, WNV, KHV, KWV>), grid,   \
                       dim3(WMV * WNV * 64), 0, s, bfp(xT), bfp(dyT),        \
                       c32.data_ptr<float>(), zp, Cin, Cout, N, Hp, Wp, OH,  \
                       OWp2, runs_total, LT, sk)
    if (k5) {
      if (CT == 32) WGR_LAUNCH(32, 4, 2, 5, 5);
      else if (CT == 16) WGR_LAUNCH(16, 4, 2, 5, 5);
      else WGR_LAUNCH(8, 4, 2, 5, 5);
    } else if (wavesn == 4) {
      if (CT == 32) WGR_LAUNCH(32, 2, 4, 3, 3);
      else if (CT == 16) WGR_LAUNCH(16, 2, 4, 3, 3);
      else WGR_LAUNCH(8, 2, 4, 3, 3);
    } else {
      if (CT == 32) WGR_LAUNCH(32, 4, 2, 3, 3);
      else if (CT == 16) WGR_LAUNCH(16, 4, 2, 3, 3);
      else WGR_LAUNCH(8, 4, 2, 3, 3);
    }
#undef WGR_LAUNCH
    run_finalize(c32, dw2d, nullptr, 0, KH * KW * Cin, Cout, sk);
    return;
  }
#define WGF_LAUNCH(CTV, WMV, WNV, KHV, KWV)                                  \
  hipLaunchKernelGGL((wgrad_fused3_kernel<CTV, WMV, WNV, KHV, KWV>), grid,   \
                     dim3(WMV * WNV * 64), 0, s, bfp(xT), bfp(dyT),          \
                     c32.data_ptr<float>(), zp, Cin, Cout, N, Hp, Wp, OH,    \
                     OWp2, int(tiles_k), sk)
  if (k5) {          // 5x5: 16ci x 32co per wave, 25 accumulator sets
    if (CT == 32) WGF_LAUNCH(32, 4, 2, 5, 5);
    else if (CT == 16) WGF_LAUNCH(16, 4, 2, 5, 5);
    else WGF_LAUNCH(8, 4, 2, 5, 5);
  } else if (wavesn == 4) {
    if (CT == 32) WGF_LAUNCH(32, 2, 4, 3, 3);
    else if (CT == 16) WGF_LAUNCH(16, 2, 4, 3, 3);
    else WGF_LAUNCH(8, 2, 4, 3, 3);
  } else {
    // Cout == 64: keep 8 waves (a 4-wave block at ~195 VGPR is 1
    // wave/SIMD — measured catastrophic on VGG-16 conv1_x) by going
    // 16ci x 32co per wave (NI=1, 72 acc regs)
    if (CT == 32) WGF_LAUNCH(32, 4, 2, 3, 3);
    else if (CT == 16) WGF_LAUNCH(16, 4, 2, 3, 3);
    else WGF_LAUNCH(8, 4, 2, 3, 3);
  }
#undef WGF_LAUNCH
  run_finalize(c32, dw2d, nullptr, 0, KH * KW * Cin, Cout, sk);
}

// glds wgrad (stride==1, Cin %% 128 == 0): substituted-reduction implicit
// GEMM with both operands staged by global_load_lds (see gemm_tile.h).
static bool wgrad_glds_ok(int Cin, int stride, int KW, int pad) {
  return stride == 1 && Cin % 128 == 0 && KW <= 8 && pad <= 8 &&
         env_int("DVGG_WGRAD_GLDS", 0);
}

static void wgrad_glds(const at::Tensor& x, const at::Tensor& dym, int KH,
                       int KW, int pad, at::Tensor& dw2d, int Cout) {
  int N = x.size(0), H = x.size(1), W = x.size(2), Cin = x.size(3);
  int OH = dym.size(1), OW = dym.size(2);
  int Kfull = KH * KW * Cin;
  int Wk = ceil8(W);
  int Kred = N * OH * Wk;
  int Hp = H + 2 * pad;
  int Wg = ceil8(W) + 16;
  auto xT = nhwc2cnhw_pad(x, Hp, Wg, pad, 8, 'g');   // cols [8, 8+W)
  int OWq = ceil8(Wk + 16) + 8;
  i64 cstride = (i64)Cout * N * OH * OWq;
  char key[96];
  snprintf(key, sizeof key, "s%d_%d_%d_%d_%d_%d", x.get_device(), KW, Cout,
           N, OH, OWq);
  auto dyTS = cached_zero_buffer(key, (i64)KW * cstride, x.options())
      .narrow(0, 0, (i64)KW * cstride);
  dim3 tgrid(N * OH, ceil_div(OW, 32), ceil_div(Cout, 32));
  hipLaunchKernelGGL(dyts_kernel, tgrid, dim3(256), 0, cur_stream(),
                     bfp(dym), bfp_mut(dyTS), N, OH, OW, Cout, OWq, KW, pad,
                     cstride);
  MDiv dOHWk{u32(OH * Wk)}, dWk{u32(Wk)};
  WgradXgGather ga{bfp(xT), Kfull, Cin, N, Hp, Wg, OH, Wk, Kred,
                   MDiv(u32(Cin)), MDiv(u32(KW)), dOHWk, dWk};
  WgradDygGather gb{bfp(dyTS), Cout, N, OH, OWq, Kred, cstride, dOHWk, dWk};
  int tiles = ceil_div(Kfull, BM) * ceil_div(Cout, BN);
  int sk = std::max(2, pick_splitk(tiles, Kred, 1024));
  auto c32 = scratch_c32("wgp", Kfull, Cout, x.options());
  launch_gemm(ga, gb, nullptr, &c32, nullptr, 0, Kfull, Cout, Kred, sk);
  run_finalize(c32, dw2d, nullptr, 0, Kfull, Cout, 1, true);
}

// --------------- ring direct conv (conv_ring.h) -----------------------
static bool conv_ring_ok(int KH, int KW, int stride, int Ci, int Co,
                         int OH) {
  return stride == 1 && KH == 3 && KW == 3 && Ci == 64 && Co % 64 == 0 &&
         OH >= 8 && env_int("DVGG_CONV_RING", 1);
}

static void conv_ring(const at::Tensor& x, const at::Tensor& wT,
                      const at::Tensor* bias, at::Tensor& y, int KH, int KW,
                      int pad, int epi) {
  int N = x.size(0), H = x.size(1), W = x.size(2);
  int OH = y.size(1), OW = y.size(2), Cout = y.size(3);
  int OWp2 = (OW + 31) / 32 * 32;
  int SEG = OWp2 / 32;
  int LT = (OH + 3) / 4;                 // TR = 4
  int runs = N * SEG;
  int blocks = runs * (Cout / 64);
  const u16* zp = zero_page(x.options());
  const u16* bp = bias ? bfp(*bias) : nullptr;
  auto s = cur_stream();
#define CRL(EPIV)                                                            \
  hipLaunchKernelGGL((conv_ring_kernel<EPIV>), dim3(blocks), dim3(512), 0,   \
                     s, bfp(x), bfp(wT), bp, bfp_mut(y), zp, N, H, W, Cout,  \
                     OH, OW, OWp2, pad, runs, LT)
  if (epi == 0) CRL(0);
  else if (epi == 1) CRL(1);
  else if (epi == 2) CRL(2);
  else CRL(3);
#undef CRL
}

// flipped+transposed weight for the ring dgrad: wFT[ci][(kh,kw,co)] =
// w[KH-1-kh][KW-1-kw][ci][co] (dx of a stride-1 conv IS a conv of dy with
// this kernel at pad' = KH-1-pad). Cached per optimizer step (kind 3).
static at::Tensor cached_wft(bool cache_w, const at::Tensor& w, int KH,
                             int KW, int Cin, int Cout) {
  return cached_weight_form(3, cache_w, w,
                            [&](c10::optional<at::Tensor> prev) {
    auto f = w.flip({0, 1}).permute({2, 0, 1, 3});   // [Cin][KH][KW][Cout]
    if (prev) {
      prev->view({Cin, KH, KW, Cout}).copy_(f);
      return *prev;
    }
    return f.contiguous().view({Cin, (i64)KH * KW * Cout});
  });
}

// Phase-interleaved NHWC transform for the small-Cin strided first
// layer (conv1 11x11/s4/C3): xP[n][hq][wq][(ph*s+pw)*C + c] =
// x[n][hq*s+ph-pad][wq*s+pw-pad][c]. The layer then lowers to a DENSE
// stride-1 conv over s*s*C = 48 phase-channels with Kq = ceil(KH/s)
// taps: K drops from 11*11*8 (channel-padded) = 968 to 3*3*48 = 432 and
// every gather is 16B-aligned. One block per (n, hq): the s source rows
// are staged in LDS, outputs written as b128.
__global__ void nhwc2phase_nhwc_kernel(const u16* __restrict__ x,
                                       u16* __restrict__ out, int N, int H,
                                       int W, int C, int s, int Hq, int Wq,
                                       int Wqp, int pad) {
  extern __shared__ u16 rows[];            // s * W * C elems
  int nh = blockIdx.x;
  int n = nh / Hq, hq = nh % Hq;
  int t = threadIdx.x;
  int rowe = W * C;
  for (int ph = 0; ph < s; ++ph) {
    int ih = hq * s + ph - pad;
    u16* dstrow = rows + ph * rowe;
    if ((unsigned)ih < (unsigned)H) {
      const u16* src = x + ((i64)n * H + ih) * (i64)rowe;
      for (int e8 = t * 8; e8 < rowe; e8 += blockDim.x * 8) {
        if (e8 + 8 <= rowe)
          *reinterpret_cast<s16x8*>(dstrow + e8) =
              *reinterpret_cast<const s16x8*>(src + e8);
        else
          for (int j = e8; j < rowe; ++j) dstrow[j] = src[j];
      }
    } else {
      for (int e = t; e < rowe; e += blockDim.x) dstrow[e] = 0;
    }
  }
  __syncthreads();
  int PC = s * s * C;                      // real phase-channels
  int PCp = (PC + 7) & ~7;                 // stored (zero-padded to 8)
  int chunks = PCp / 8;
  // fixed chunk per thread: the 8 (ph,pw,c) decodes (v_rcp div chains
  // for C=3) hoist out of the position loop
  int wstride = blockDim.x / chunks;
  if (t >= chunks * wstride) return;       // (after the barrier above)
  int k8 = (t % chunks) * 8;
  int ofs[8], pw8[8];
  bool inr[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    int pcc = k8 + j;
    int c = pcc % C, pw = (pcc / C) % s, ph = pcc / (C * s);
    ofs[j] = ph * W * C + c;
    pw8[j] = pw - pad;
    inr[j] = pcc < PC;
  }
  for (int wq = t / chunks; wq < Wq; wq += wstride) {
    union { s16x8 v; u16 e[8]; } pk;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int iw = wq * s + pw8[j];
      pk.e[j] = (inr[j] && (unsigned)iw < (unsigned)W)
          ? rows[ofs[j] + iw * C] : u16(0);
    }
    *reinterpret_cast<s16x8*>(
        out + (((i64)n * Hq + hq) * Wqp + wq) * PCp + k8) = pk.v;
  }
}

// ------------------------------------------------------------------ conv
static void conv_out_shape(int H, int W, int KH, int KW, int stride, int pad,
                           int& OH, int& OW) {
  OH = (H + 2 * pad - KH) / stride + 1;
  OW = (W + 2 * pad - KW) / stride + 1;
}

at::Tensor conv2d_fwd(const at::Tensor& x, const at::Tensor& w,
                      const c10::optional<at::Tensor>& bias, int64_t stride_,
                      int64_t pad_, bool relu, bool cache_w) {
  check_bf16(x, "x"); check_bf16(w, "w");
  int stride = int(stride_), pad = int(pad_);
  int N = x.size(0), H = x.size(1), W = x.size(2), Cin = x.size(3);
  int KH = w.size(0), KW = w.size(1), Cout = w.size(3);
  TORCH_CHECK(w.size(2) == Cin, "conv weight Cin mismatch");
  int OH, OW;
  conv_out_shape(H, W, KH, KW, stride, pad, OH, OW);
  int M = N * OH * OW, Kfull = KH * KW * Cin;
  if (KH == 1 && KW == 1 && stride == 1 && pad == 0 && Cin % 8 == 0) {
    // 1x1/s1 conv IS the FC GEMM
    auto y2 = linear_fwd(x.view({M, Cin}), w.view({Cin, Cout}), bias,
                         relu, cache_w);
    return y2.view({N, OH, OW, Cout});
  }
  auto y = at::empty({N, OH, OW, Cout}, x.options());
  auto y2d = y.view({M, Cout});
  int epi = epi_of(bias.has_value(), relu);

  if (Cin < 8) {
    // conv1 (11x11/s4/C3): phase-plane lowering — a DENSE stride-1 conv
    // over s*s*C = 48 phase-channels with ceil(KH/s) = 3 taps (K 968 ->
    // 432 vs the channel-padded form, all gathers 16B-aligned)
    int PC = stride * stride * Cin;
    int PCp = ceil8(PC);
    if (stride >= 2 && KH == KW && env_int("DVGG_CONV1_PHASE", 1)) {
      int Kq = (KH + stride - 1) / stride;
      int Hq = OH + Kq - 1, Wq = OW + Kq - 1;
      int Wqp = Wq;  // dense layout, no pad needed (pad folded into xP)
      auto xP = at::empty({N, Hq, Wqp, PCp}, x.options());
      TORCH_CHECK(stride * W * Cin <= 16384, "phase rows exceed LDS");
      hipLaunchKernelGGL(nhwc2phase_nhwc_kernel, dim3(N * Hq), dim3(256),
                         stride * W * Cin * sizeof(u16), cur_stream(),
                         bfp(x), bfp_mut(xP), N, H, W, Cin, stride, Hq, Wq,
                         Wqp, pad);
      // wP[q][r][(ph,pw,c)][co] = w[s q + ph][s r + pw][c][co], zero
      // padded beyond KH; cached per optimizer step in its final layout
      auto wP = cached_weight_form(4, cache_w, w,
                                   [&](c10::optional<at::Tensor> prev) {
        auto t6 = at::zeros({Kq * stride, Kq * stride, Cin, Cout},
                            w.options());
        t6.narrow(0, 0, KH).narrow(1, 0, KW).copy_(w);
        auto perm = t6.view({Kq, stride, Kq, stride, Cin, Cout})
            .permute({0, 2, 1, 3, 4, 5});
        if (prev) {
          prev->view({Kq, Kq, (i64)PCp, Cout}).narrow(2, 0, PC)
              .copy_(perm.reshape({Kq, Kq, (i64)PC, Cout}));
          return *prev;
        }
        auto t = at::zeros({Kq, Kq, (i64)PCp, Cout}, w.options());
        t.narrow(2, 0, PC).copy_(perm.reshape({Kq, Kq, (i64)PC, Cout}));
        return t;
      });
      return conv2d_fwd(xP, wP, bias, 1, 0, relu, cache_w);
    }
    // ResNet-stem-style fallback: pad channels to 8 and take the
    // vector-gather glds path; zero channels contribute nothing
    auto x8 = pad_channels8(x);
    auto w8 = cached_weight_form(2, cache_w, w, [&](c10::optional<at::Tensor> prev) {
      auto t = prev ? *prev : at::zeros({KH, KW, 8, Cout}, w.options());
      t.narrow(2, 0, Cin).copy_(w);
      return t;
    });
    return conv2d_fwd(x8, w8, bias, stride_, pad_, relu, cache_w);
  }
  TORCH_CHECK(Cin % 8 == 0, "conv fwd needs Cin % 8 == 0 (or Cin < 8)");
  auto w2d = w.view({Kfull, Cout});
  auto wT = cached_weight_form(0, cache_w, w, [&](c10::optional<at::Tensor> prev) {
    if (prev) { transpose2d_into(w2d, *prev); return *prev; }
    return transpose2d(w2d);
  });  // [Cout][K]
  if (conv_ring_ok(KH, KW, stride, Cin, Cout, OH)) {
    conv_ring(x, wT, bias ? &*bias : nullptr, y, KH, KW, pad, epi);
    return y;
  }
  ConvFwdGather ga{bfp(x), M, H, W, Cin, stride, pad,
                   MDiv(u32(OH * OW)), MDiv(u32(OW)), MDiv(u32(Cin)),
                   MDiv(u32(KW))};
  PlainGather gb{bfp(wT), Cout, Kfull, Kfull};
  int tiles = ceil_div(M, BM) * ceil_div(Cout, BN);
  int sk = pick_splitk(tiles, Kfull);
  if (sk > 1) {
    auto c32 = scratch_c32("cvf", M, Cout, x.options());
    launch_gemm(ga, gb, nullptr, &c32, nullptr, 0, M, Cout, Kfull, sk);
    run_finalize(c32, y, bias ? &*bias : nullptr, epi, M, Cout, 1, true);
  } else {
    launch_gemm(ga, gb, &y, nullptr, bias ? &*bias : nullptr, epi, M, Cout,
                Kfull, 1);
  }
  return y;
}

at::Tensor conv2d_dgrad(const at::Tensor& dy, const at::Tensor& w, int64_t H_,
                        int64_t W_, int64_t stride_, int64_t pad_,
                        bool cache_w) {
  check_bf16(dy, "dy"); check_bf16(w, "w");
  int H = int(H_), W = int(W_), stride = int(stride_), pad = int(pad_);
  int N = dy.size(0), OH = dy.size(1), OW = dy.size(2), Cout = dy.size(3);
  int KH = w.size(0), KW = w.size(1), Cin = w.size(2);
  TORCH_CHECK(Cout % 8 == 0, "dgrad needs Cout % 8 == 0");
  TORCH_CHECK(Cin % 8 == 0,
              "dgrad for Cin<8 first layers is unused by design");
  int M = N * H * W, K = KH * KW * Cout;
  auto dx = at::empty({N, H, W, Cin}, dy.options());
  if (conv_ring_ok(KH, KW, stride, Cout, Cin, H)) {
    auto wFT = cached_wft(cache_w, w, KH, KW, Cin, Cout);
    conv_ring(dy, wFT, nullptr, dx, KH, KW, KH - 1 - pad, 0);
    return dx;
  }
  // Bt[ci][(kh,kw,co)] = w[kh][kw][ci][co]: permute HWIO -> [Cin][KhKw*Cout]
  // KH*KW==1: [1][Cin][Cout] permuted is w itself — a free view, and no
  // cached copy to refresh every step (was a DtoD blit per layer per step)
  auto wD = (KH * KW == 1)
      ? w.view({Cin, (i64)Cout})
      : cached_weight_form(1, cache_w, w,
                           [&](c10::optional<at::Tensor> prev) {
          auto p = w.view({KH * KW, Cin, Cout}).permute({1, 0, 2});
          if (prev) { prev->view({Cin, KH * KW, Cout}).copy_(p); return *prev; }
          return p.contiguous().view({Cin, (i64)KH * KW * Cout});
        });
  TORCH_CHECK(stride == 1 || stride == 2, "dgrad supports stride 1/2");
  ConvDgradGather ga{bfp(dy), M, H, W, Cout, OH, OW, stride, pad,
                     MDiv(u32(H * W)), MDiv(u32(W)), MDiv(u32(Cout)),
                     MDiv(u32(KW))};
  PlainGather gb{bfp(wD), Cin, K, K};
  int tiles = ceil_div(M, BM) * ceil_div(Cin, BN);
  int sk = pick_splitk(tiles, K);
  if (sk > 1) {
    auto c32 = scratch_c32("cvd", M, Cin, dy.options());
    launch_gemm(ga, gb, nullptr, &c32, nullptr, 0, M, Cin, K, sk);
    run_finalize(c32, dx, nullptr, 0, M, Cin, 1, true);
  } else {
    launch_gemm(ga, gb, &dx, nullptr, nullptr, 0, M, Cin, K, 1);
  }
  return dx;
}

at::Tensor conv2d_wgrad(const at::Tensor& dy, const at::Tensor& x, int64_t KH_,
                        int64_t KW_, int64_t stride_, int64_t pad_) {
  check_bf16(dy, "dy"); check_bf16(x, "x");
  int KH = int(KH_), KW = int(KW_), stride = int(stride_), pad = int(pad_);
  int N = x.size(0), H = x.size(1), W = x.size(2), Cin = x.size(3);
  int OH = dy.size(1), OW = dy.size(2), Cout = dy.size(3);
  int Kfull = KH * KW * Cin;
  auto dw = at::empty({KH, KW, Cin, Cout}, x.options());
  auto dw2d = dw.view({Kfull, Cout});

  if (wgrad_glds_ok(Cin, stride, KW, pad)) {
    auto dyc = dy.contiguous();
    wgrad_glds(x, dyc, KH, KW, pad, dw2d, Cout);
    return dw;
  }
  if (KH == 1 && KW == 1 && pad == 0 && Cin % 8 == 0 && stride > 1) {
    // strided 1x1 (ResNet downsample): TN GEMM over x sampled at
    // (s*oh, s*ow) — see the conv2d_bwd fast path
    int M1 = N * OH * OW;
    int Mred = ((M1 + BK - 1) / BK) * BK;
    auto dy2 = dy.contiguous().view({M1, Cout});
    StridedRowGather ga{bfp(x), M1, Cin, H, W, stride,
                        MDiv(u32(OH * OW)), MDiv(u32(OW))};
    PlainGather gb{bfp(dy2), M1, Cout, Cout};
    int tiles1 = ceil_div(Cin, BM) * ceil_div(Cout, BN);
    int sk1 = pick_splitk(tiles1, Mred);
    auto dwt = dw2d.view({Cin, Cout});
    if (sk1 > 1) {
      auto c32w = scratch_c32("w11", Cin, Cout, x.options());
      launch_gemm_tn(ga, gb, nullptr, &c32w, Cin, Cout, Mred, sk1);
      run_finalize(c32w, dwt, nullptr, 0, Cin, Cout, 1, true);
    } else {
      launch_gemm_tn(ga, gb, &dwt, nullptr, Cin, Cout, Mred, 1);
    }
    return dw;
  }
  // Padding Cin<8 to 8 was tried and measured 3.4x SLOWER for wgrad
  // (zero-channel waste x scalar stride-4 staging); the direct NHWC
  // gather below stays for Cin < 8.
  bool wfused = wgrad_fused_ok(KH, KW, stride, Cin, Cout);
  int OWp = wfused ? owp2_of(OW) : ceil8(OW);
  int Kred = N * OH * OWp;
  auto dyT = nhwc2cnhw_pad(dy, OH, OWp, 0, 0, 'd');  // [Cout][N][OH][OWp]
  if (wfused) {
    int Hp = H + 2 * pad;
    int Wp = ceil8(std::max(W + 2 * pad, (OWp - 1) * stride + KW));
    auto xT = nhwc2cnhw_pad(x, Hp, Wp, pad, pad, 'x');
    wgrad_fused(xT, dyT, dw2d, KH, KW, Cin, Cout, N, Hp, Wp, OH, OWp);
    return dw;
  }
  WgradDyGather gb{bfp(dyT), Cout, N, OH, OWp, Kred};
  int tiles = ceil_div(Kfull, BM) * ceil_div(Cout, BN);
  int sk = std::max(2, pick_splitk(tiles, Kred, 1024));
  auto c32 = scratch_c32("wgt", Kfull, Cout, x.options());
  bool phase_ok = Cin % 8 != 0 && (stride == 2 || stride == 4) &&
                  ((KW - 1) / stride) < 8 && W * Cin <= 4096 &&
                  env_int("DVGG_WGRAD_PHASE", 1);
  if (phase_ok) {
    // stride-phase planes: the A gather becomes stride-1 funnel loads
    // (padding is folded into the plane layout as a row/col offset)
    int phl = stride == 4 ? 2 : 1;
    int KHq = (KH - 1) / stride, KWq = (KW - 1) / stride;
    int Hq = std::max(ceil_div(H + 2 * pad, stride), OH + KHq + 1);
    int Wqp = ceil8(std::max(ceil_div(W + 2 * pad, stride), OWp + KWq));
    auto xP = nhwc2phase(x, stride, Hq, Wqp, pad);
    WgradXGather ga{bfp(xP), Kfull, Cin, N, Hq, Wqp, OH, OWp, 1, Kred,
                    MDiv(u32(Cin)), MDiv(u32(KW)), MDiv(u32(OH * OWp)),
                    MDiv(u32(OWp)), phl};
    launch_gemm(ga, gb, nullptr, &c32, nullptr, 0, Kfull, Cout, Kred, sk);
  } else if (Cin % 8 != 0) {
    WgradXDirectGather ga{bfp(x), Kfull, Cin, N, H, W, OH, OWp, stride,
                          pad, Kred, MDiv(u32(Cin)), MDiv(u32(KW)),
                          MDiv(u32(OH * OWp)), MDiv(u32(OWp))};
    launch_gemm(ga, gb, nullptr, &c32, nullptr, 0, Kfull, Cout, Kred, sk);
  } else {
    int Hp = H + 2 * pad;
    int Wp = ceil8(std::max(W + 2 * pad, (OWp - 1) * stride + KW));
    auto xT = nhwc2cnhw_pad(x, Hp, Wp, pad, pad, 'x');  // [Cin][N][Hp][Wp]
    WgradXGather ga{bfp(xT), Kfull, Cin, N, Hp, Wp, OH, OWp, stride, Kred,
                    MDiv(u32(Cin)), MDiv(u32(KW)), MDiv(u32(OH * OWp)),
                    MDiv(u32(OWp)), 0};
    launch_gemm_wgrad(ga, gb, &c32, Kfull, Cout, Kred, sk);
  }
  run_finalize(c32, dw2d, nullptr, 0, Kfull, Cout, 1, true);
  return dw;
}

// Fused conv backward (SURVEY.md C-path): preamble -> dgrad GEMM -> wgrad
// GEMM. Returns {dx (or empty), dw, db (or empty)}.
std::vector<at::Tensor> conv2d_bwd(const at::Tensor& dy, const at::Tensor& y,
                                   const at::Tensor& x, const at::Tensor& w,
                                   int64_t stride_, int64_t pad_, bool relu,
                                   bool need_dx, bool need_db,
                                   const c10::optional<at::Tensor>& dw_out,
                                   const c10::optional<at::Tensor>& db_out,
                                   bool cache_w) {
  check_bf16(dy, "dy"); check_bf16(x, "x"); check_bf16(w, "w");
  int stride = int(stride_), pad = int(pad_);
  int N = x.size(0), H = x.size(1), W = x.size(2), Cin = x.size(3);
  int KH = w.size(0), KW = w.size(1), Cout = w.size(3);
  int OH = dy.size(1), OW = dy.size(2);
  bool wfused = wgrad_fused_ok(KH, KW, stride, Cin, Cout);
  int OWp = wfused ? owp2_of(OW) : ceil8(OW);
  int M = N * OH * OW, Kfull = KH * KW * Cin;
  auto s = cur_stream();

  if (KH == 1 && KW == 1 && pad == 0 && Cin % 8 == 0) {
    // 1x1 conv backward == FC backward on flattened spatial dims (no
    // transposed dy staging). stride>1 (ResNet downsample): the wgrad
    // reduction samples x at (s*oh, s*ow) via StridedRowGather — the
    // implicit-GEMM wgrad path measured ~5x slower on these; dgrad
    // keeps the implicit path (dx scatter is strided).
    auto dy2 = dy.view({M, Cout});
    if (relu) dy2 = relu_bwd(dy2, y.view({M, Cout}));
    at::Tensor dx1;
    if (need_dx) {
      if (stride == 1)
        dx1 = linear_dgrad(dy2, w.view({Cin, Cout})).view({N, H, W, Cin});
      else
        dx1 = conv2d_dgrad(dy2.view({N, OH, OW, Cout}), w, H, W, stride,
                           pad, cache_w);
    }
    at::Tensor dw1;
    if (stride == 1) {
      dw1 = linear_wgrad(x.view({M, Cin}), dy2, dw_out)
          .view({1, 1, Cin, Cout});
    } else {
      auto dwt = dw_out ? dw_out->view({Cin, Cout})
                        : at::empty({Cin, Cout}, x.options());
      int Mred = ((M + BK - 1) / BK) * BK;
      StridedRowGather ga{bfp(x), M, Cin, H, W, stride,
                          MDiv(u32(OH * OW)), MDiv(u32(OW))};
      PlainGather gb{bfp(dy2), M, Cout, Cout};
      int tiles = ceil_div(Cin, BM) * ceil_div(Cout, BN);
      int sk = pick_splitk(tiles, Mred);
      if (sk > 1) {
        auto c32w = scratch_c32("w11b", Cin, Cout, x.options());
        launch_gemm_tn(ga, gb, nullptr, &c32w, Cin, Cout, Mred, sk);
        run_finalize(c32w, dwt, nullptr, 0, Cin, Cout, 1, true);
      } else {
        launch_gemm_tn(ga, gb, &dwt, nullptr, Cin, Cout, Mred, 1);
      }
      dw1 = dwt.view({1, 1, Cin, Cout});
    }
    at::Tensor db1;
    if (need_db) db1 = bias_grad(dy2, db_out);
    return {dx1, dw1, db1};
  }

  // ---- preamble: masked dy + dyTpad + db32 in one pass over dy
  // (Cin < 8 falls through: wgrad below uses the direct NHWC gather)
  i64 tnumel = (i64)Cout * N * OH * OWp;
  char key[96];
  snprintf(key, sizeof key, "b%d_%d_%d_%d_%d", dy.get_device(), Cout, N, OH,
           OWp);
  auto dyT = cached_zero_buffer(key, tnumel, dy.options())
      .narrow(0, 0, tnumel).view({Cout, N, OH, OWp});
  // dym must be masked whenever a downstream consumer reads it: dgrad
  // (need_dx) or the glds wgrad path (which consumes dym, not dyT)
  bool glds_path = wgrad_glds_ok(Cin, stride, KW, pad);
  bool write_dym = relu && (need_dx || glds_path);
  at::Tensor dym = write_dym ? at::empty_like(dy) : dy;
  auto db32 = need_db ? scratch_c32("dbs", 256, Cout, dy.options())
                      : at::Tensor();  // bias_slot_reduce re-zeros it
  const u16* yp = relu ? bfp(y) : nullptr;
  float* dbp = need_db ? db32.data_ptr<float>() : nullptr;
  u16* dymp = write_dym ? bfp_mut(dym) : nullptr;
  u16* dyTp = bfp_mut(dyT);
  bool vec8 = (Cout % 8 == 0) && (OWp % 8 == 0);
  dim3 pgrid(N * OH, ceil_div(vec8 ? OWp : OW, vec8 ? 64 : 32),
             ceil_div(Cout, vec8 ? 64 : 32));
#define DVGG_PRE_LAUNCH(R, DM, DBF)                                          \
  do {                                                                       \
    if (vec8)                                                                \
      hipLaunchKernelGGL((bwd_preamble8_kernel<R, DM, DBF>), pgrid,          \
                         dim3(256), 0, s, bfp(dy), yp, dymp, dyTp, dbp, N,   \
                         OH, OW, Cout, OWp);                                 \
    else                                                                     \
      hipLaunchKernelGGL((bwd_preamble_kernel<R, DM, DBF>), pgrid,           \
                         dim3(256), 0, s, bfp(dy), yp, dymp, dyTp, dbp, N,   \
                         OH, OW, Cout, OWp);                                 \
  } while (0)
  if (relu && write_dym && need_db)    DVGG_PRE_LAUNCH(true, true, true);
  else if (relu && write_dym)          DVGG_PRE_LAUNCH(true, true, false);
  else if (relu && need_db)            DVGG_PRE_LAUNCH(true, false, true);
  else if (relu)                       DVGG_PRE_LAUNCH(true, false, false);
  else if (need_db)                    DVGG_PRE_LAUNCH(false, false, true);
  else                                 DVGG_PRE_LAUNCH(false, false, false);
#undef DVGG_PRE_LAUNCH
  // note: when relu && !write_dym, dym (== dy) is unused downstream;
  // the masked values only exist in dyT. wgrad/db see masked dy either way.

  // ---- dgrad
  at::Tensor dx;
  if (need_dx) {
    TORCH_CHECK(Cout % 8 == 0 && Cin % 8 == 0, "dgrad channel alignment");
    int K = KH * KW * Cout;
    dx = at::empty({N, H, W, Cin}, dy.options());
    if (conv_ring_ok(KH, KW, stride, Cout, Cin, H)) {
      auto wFT = cached_wft(cache_w, w, KH, KW, Cin, Cout);
      conv_ring(dym.contiguous(), wFT, nullptr, dx, KH, KW, KH - 1 - pad, 0);
      goto dgrad_done;
    }
    {
    auto wD = (KH * KW == 1)  // free view of w; nothing to cache/refresh
        ? w.view({Cin, (i64)Cout})
        : cached_weight_form(1, cache_w, w,
                             [&](c10::optional<at::Tensor> prev) {
            auto p = w.view({KH * KW, Cin, Cout}).permute({1, 0, 2});
            if (prev) {
              prev->view({Cin, KH * KW, Cout}).copy_(p);
              return *prev;
            }
            return p.contiguous().view({Cin, (i64)KH * KW * Cout});
          });
    TORCH_CHECK(stride == 1 || stride == 2, "dgrad supports stride 1/2");
    int Mx = N * H * W;
    ConvDgradGather ga{bfp(dym), Mx, H, W, Cout, OH, OW, stride, pad,
                       MDiv(u32(H * W)), MDiv(u32(W)), MDiv(u32(Cout)),
                       MDiv(u32(KW))};
    PlainGather gb{bfp(wD), Cin, K, K};
    int tiles = ceil_div(Mx, BM) * ceil_div(Cin, BN);
    int sk = pick_splitk(tiles, K);
    if (sk > 1) {
      auto c32 = scratch_c32("d11", Mx, Cin, dy.options());
      launch_gemm(ga, gb, nullptr, &c32, nullptr, 0, Mx, Cin, K, sk);
      run_finalize(c32, dx, nullptr, 0, Mx, Cin, 1, true);
    } else {
      launch_gemm(ga, gb, &dx, nullptr, nullptr, 0, Mx, Cin, K, 1);
    }
    }
    dgrad_done:;
  }

  // ---- wgrad
  auto dw = dw_out ? *dw_out : at::empty({KH, KW, Cin, Cout}, x.options());
  auto dw2d = dw.view({Kfull, Cout});
  if (glds_path) {
    // dym is ReLU-masked whenever relu (write_dym covers this path); dyT
    // from the preamble is unused here
    auto dymc = dym.contiguous();
    wgrad_glds(x, dymc, KH, KW, pad, dw2d, Cout);
    at::Tensor db;
    if (need_db) {
      db = db_out ? *db_out : at::empty({Cout}, dy.options());
      hipLaunchKernelGGL(bias_slot_reduce_kernel, dim3(ceil_div(Cout, 256)),
                         dim3(256), 0, s, db32.data_ptr<float>(),
                         bfp_mut(db), Cout);
    }
    return {need_dx ? dx : at::Tensor(), dw, need_db ? db : at::Tensor()};
  }
  if (wfused) {
    int Hp = H + 2 * pad;
    int Wp = ceil8(std::max(W + 2 * pad, (OWp - 1) * stride + KW));
    auto xT = nhwc2cnhw_pad(x, Hp, Wp, pad, pad, 'x');
    wgrad_fused(xT, dyT, dw2d, KH, KW, Cin, Cout, N, Hp, Wp, OH, OWp);
    at::Tensor db;
    if (need_db) {
      db = db_out ? *db_out : at::empty({Cout}, dy.options());
      hipLaunchKernelGGL(bias_slot_reduce_kernel, dim3(ceil_div(Cout, 256)),
                         dim3(256), 0, s, db32.data_ptr<float>(),
                         bfp_mut(db), Cout);
    }
    return {need_dx ? dx : at::Tensor(), dw, need_db ? db : at::Tensor()};
  }
  int Kred = N * OH * OWp;
  WgradDyGather gb{bfp(dyT), Cout, N, OH, OWp, Kred};
  int tiles = ceil_div(Kfull, BM) * ceil_div(Cout, BN);
  int sk = std::max(2, pick_splitk(tiles, Kred, 1024));
  auto c32 = scratch_c32("wgt", Kfull, Cout, x.options());
  bool phase_ok = Cin % 8 != 0 && (stride == 2 || stride == 4) &&
                  ((KW - 1) / stride) < 8 && W * Cin <= 4096 &&
                  env_int("DVGG_WGRAD_PHASE", 1);
  if (phase_ok) {
    // stride-phase planes: the A gather becomes stride-1 funnel loads
    // (padding is folded into the plane layout as a row/col offset)
    int phl = stride == 4 ? 2 : 1;
    int KHq = (KH - 1) / stride, KWq = (KW - 1) / stride;
    int Hq = std::max(ceil_div(H + 2 * pad, stride), OH + KHq + 1);
    int Wqp = ceil8(std::max(ceil_div(W + 2 * pad, stride), OWp + KWq));
    auto xP = nhwc2phase(x, stride, Hq, Wqp, pad);
    WgradXGather ga{bfp(xP), Kfull, Cin, N, Hq, Wqp, OH, OWp, 1, Kred,
                    MDiv(u32(Cin)), MDiv(u32(KW)), MDiv(u32(OH * OWp)),
                    MDiv(u32(OWp)), phl};
    launch_gemm(ga, gb, nullptr, &c32, nullptr, 0, Kfull, Cout, Kred, sk);
  } else if (Cin % 8 != 0) {
    WgradXDirectGather ga{bfp(x), Kfull, Cin, N, H, W, OH, OWp, stride,
                          pad, Kred, MDiv(u32(Cin)), MDiv(u32(KW)),
                          MDiv(u32(OH * OWp)), MDiv(u32(OWp))};
    launch_gemm(ga, gb, nullptr, &c32, nullptr, 0, Kfull, Cout, Kred, sk);
  } else {
    int Hp = H + 2 * pad;
    int Wp = ceil8(std::max(W + 2 * pad, (OWp - 1) * stride + KW));
    auto xT = nhwc2cnhw_pad(x, Hp, Wp, pad, pad, 'x');
    WgradXGather ga{bfp(xT), Kfull, Cin, N, Hp, Wp, OH, OWp, stride, Kred,
                    MDiv(u32(Cin)), MDiv(u32(KW)), MDiv(u32(OH * OWp)),
                    MDiv(u32(OWp)), 0};
    launch_gemm_wgrad(ga, gb, &c32, Kfull, Cout, Kred, sk);
  }
  run_finalize(c32, dw2d, nullptr, 0, Kfull, Cout, 1, true);

  at::Tensor db;
  if (need_db) {
    db = db_out ? *db_out : at::empty({Cout}, dy.options());
    hipLaunchKernelGGL(bias_slot_reduce_kernel, dim3(ceil_div(Cout, 256)),
                       dim3(256), 0, s, db32.data_ptr<float>(), bfp_mut(db),
                       Cout);
  }
  return {need_dx ? dx : at::Tensor(), dw, need_db ? db : at::Tensor()};
}

}  // namespace dvgg
