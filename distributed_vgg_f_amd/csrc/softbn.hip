// Fused softmax + cross-entropy and training-mode
// BatchNorm fwd/bwd (K17, ResNet-50 config). NHWC; channels innermost.

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

DEV float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_down(v, off, 64));
  return __shfl(v, 0, 64);
}
DEV float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return __shfl(v, 0, 64);
}

// -------------------------------------------------- softmax + xent (K13)
// One 256-thread block per row; computes per-row loss (fp32) AND
// dlogits = softmax - onehot (bf16) in a single pass structure.
__global__ void softmax_xent_kernel(const u16* __restrict__ logits,
                                    const long* __restrict__ labels,
                                    float* __restrict__ loss,
                                    u16* __restrict__ dlogits, int M, int C) {
  __shared__ float red[8];
  int m = blockIdx.x;
  if (m >= M) return;
  const u16* row = logits + (i64)m * C;
  u16* drow = dlogits + (i64)m * C;
  int t = threadIdx.x, lane = t & 63, wave = t >> 6;

  float mx = -3.4e38f;
  for (int c = t; c < C; c += blockDim.x) mx = fmaxf(mx, bf2f(row[c]));
  mx = wave_reduce_max(mx);
  if (lane == 0) red[wave] = mx;
  __syncthreads();
  mx = fmaxf(fmaxf(red[0], red[1]), fmaxf(red[2], red[3]));

  float sum = 0.f;
  for (int c = t; c < C; c += blockDim.x) sum += __expf(bf2f(row[c]) - mx);
  sum = wave_reduce_sum(sum);
  if (lane == 0) red[4 + wave] = sum;
  __syncthreads();
  sum = red[4] + red[5] + red[6] + red[7];
  float inv = 1.f / sum;

  long lbl = labels[m];
  for (int c = t; c < C; c += blockDim.x) {
    float p = __expf(bf2f(row[c]) - mx) * inv;
    drow[c] = f2bf(p - (c == lbl ? 1.f : 0.f));
  }
  if (t == 0) loss[m] = logf(sum) + mx - bf2f(row[lbl]);
}

std::vector<at::Tensor> softmax_xent_fwd(const at::Tensor& logits,
                                         const at::Tensor& labels) {
  check_bf16(logits, "logits");
  TORCH_CHECK(labels.scalar_type() == at::kLong && labels.is_cuda());
  int M = logits.size(0), C = logits.size(1);
  auto loss = at::empty({M}, logits.options().dtype(at::kFloat));
  auto dlogits = at::empty_like(logits);
  hipLaunchKernelGGL(softmax_xent_kernel, dim3(M), dim3(256), 0, cur_stream(),
                     bfp(logits), labels.data_ptr<long>(),
                     loss.data_ptr<float>(), bfp_mut(dlogits), M, C);
  return {loss, dlogits};
}

// --------------------------------------------------------- BN fwd (K17)
// Pass 1: per-channel sum / sumsq via block partials + fp32 atomics.
// Pass 2: normalize (+affine, optional fused ReLU); saves mean/rstd fp32.
// partials land in 256 slots per channel (slot = blockIdx.y & 255):
// direct atomics onto C floats from thousands of blocks serialize.
// C >= 256: thread per channel (whole block live). C < 256 (power of
// 2): threads fold into W = 256/C row sub-stripes per channel — without
// this 3 of 4 waves exited instantly on C=64/128 layers and the reduce
// ran ~4x short of the memory-level parallelism the HBM latency needs.
// Sub-stripe partials combine through LDS, one atomic per channel.
__global__ void bn_reduce_kernel(const u16* __restrict__ x,
                                 float* __restrict__ sum,
                                 float* __restrict__ sumsq, i64 R, int C,
                                 int rows_per_block) {
  __shared__ float ls[256], lq[256];
  int t = threadIdx.x;
  int W = 1, c = blockIdx.x * blockDim.x + t, w = 0;
  if (C < 256 && (C & (C - 1)) == 0) {
    W = 256 / C; c = t & (C - 1); w = t / C;
  } else if (c >= C) {
    return;
  }
  i64 r0 = (i64)blockIdx.y * rows_per_block;
  i64 r1 = min(R, r0 + rows_per_block);
  float s0 = 0.f, s1 = 0.f, q0 = 0.f, q1 = 0.f;
  i64 r = r0 + 2 * w;                     // stripes of 2 rows, stride 2W
  for (; r + 2 <= r1; r += 2 * W) {
    float v0 = bf2f(x[r * C + c]), v1 = bf2f(x[(r + 1) * C + c]);
    s0 += v0; q0 += v0 * v0; s1 += v1; q1 += v1 * v1;
  }
  if (r < r1) { float v = bf2f(x[r * C + c]); s0 += v; q0 += v * v; }
  float sa = s0 + s1, qa = q0 + q1;
  if (W > 1) {
    ls[t] = sa; lq[t] = qa;
    __syncthreads();
    if (t >= C) return;
    for (int j = 1; j < W; ++j) { sa += ls[t + j * C]; qa += lq[t + j * C]; }
  }
  int slot = blockIdx.y & 255;
  atomicAdd(&sum[(i64)slot * C + c], sa);
  atomicAdd(&sumsq[(i64)slot * C + c], qa);
}

__global__ void bn_stats_kernel(float* __restrict__ sum,
                                float* __restrict__ sumsq,
                                const u16* __restrict__ gamma,
                                const u16* __restrict__ beta,
                                float* __restrict__ mean,
                                float* __restrict__ rstd,
                                float* __restrict__ scale,
                                float* __restrict__ shift,
                                float* __restrict__ run_mean,
                                float* __restrict__ run_var, i64 R, int C,
                                float eps, float momentum) {
  __shared__ float ls[256], lq[256];
  int tid = threadIdx.x;
  int W = 1, c = blockIdx.x * blockDim.x + tid, w = 0;
  if (C < 256 && (C & (C - 1)) == 0) {    // see bn_reduce_kernel note
    W = 256 / C; c = tid & (C - 1); w = tid / C;
  } else if (c >= C) {
    return;
  }
  float sa = 0.f, qa = 0.f;
#pragma unroll 8
  for (int t = w; t < 256; t += W) {
    sa += sum[(i64)t * C + c];
    qa += sumsq[(i64)t * C + c];
  }
#pragma unroll 8
  for (int t = w; t < 256; t += W) {  // cached_part contract: leave zeroed
    sum[(i64)t * C + c] = 0.f;
    sumsq[(i64)t * C + c] = 0.f;
  }
  if (W > 1) {
    ls[tid] = sa; lq[tid] = qa;
    __syncthreads();
    if (tid >= C) return;
    for (int j = 1; j < W; ++j) { sa += ls[tid + j * C]; qa += lq[tid + j * C]; }
  }
  float m = sa / float(R);
  float v = fmaxf(qa / float(R) - m * m, 0.f);
  mean[c] = m;
  float rs = rsqrtf(v + eps);
  rstd[c] = rs;
  // y = x * scale + shift (the apply kernel is pure vector FMA)
  float sc = rs * bf2f(gamma[c]);
  scale[c] = sc;
  shift[c] = bf2f(beta[c]) - m * sc;
  if (run_mean) {  // running-stat update fused in (eval-mode parity)
    run_mean[c] = momentum * run_mean[c] + (1.f - momentum) * m;
    run_var[c] = momentum * run_var[c] + (1.f - momentum) * v;
  }
}

template <bool RELU, bool RES>
__global__ void bn_apply_kernel(const u16* __restrict__ x,
                                const u16* __restrict__ res,
                                const float4* __restrict__ scale4,
                                const float4* __restrict__ shift4,
                                u16* __restrict__ y, i64 total8, int C) {
  i64 id = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  if (id >= total8) return;
  int nch = C / 8;
  int c8 = int(id % nch) * 8;
  i64 o = (id / nch) * C + c8;
  s16x8 xv = *reinterpret_cast<const s16x8*>(x + o);
  s16x8 rv{};
  if (RES) rv = *reinterpret_cast<const s16x8*>(res + o);
  float4 sA = scale4[c8 / 4], sB = scale4[c8 / 4 + 1];
  float4 hA = shift4[c8 / 4], hB = shift4[c8 / 4 + 1];
  float sc[8] = {sA.x, sA.y, sA.z, sA.w, sB.x, sB.y, sB.z, sB.w};
  float sh[8] = {hA.x, hA.y, hA.z, hA.w, hB.x, hB.y, hB.z, hB.w};
  s16x8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float v = bf2f(u16(xv[j])) * sc[j] + sh[j];
    if (RES) v += bf2f(u16(rv[j]));     // K18 fused: the residual add
    if (RELU) v = v > 0.f ? v : 0.f;
    out[j] = short(f2bf(v));
  }
  *reinterpret_cast<s16x8*>(y + o) = out;
}


// Cached {2,256,C} fp32 partial-slot buffers for the BN reduces: the
// stats kernel re-zeros the slots after summing them, so the per-layer
// at::zeros Fill launch (2/layer/step on ResNet-50 = ~106 launch-latency
// fills per step) disappears. tag 0 = fwd sum/sumsq, 1 = bwd dy/dyx.
static at::Tensor cached_part(int tag, int C, const at::TensorOptions& f32) {
  static std::unordered_map<i64, at::Tensor> cache;
  i64 key = (i64)tag << 32 | (u32)C;
  auto it = cache.find(key);
  if (it == cache.end())
    it = cache.emplace(key, at::zeros({2, 256, C}, f32)).first;
  return it->second;
}

std::vector<at::Tensor> bn_fwd(const at::Tensor& x, const at::Tensor& gamma,
                               const at::Tensor& beta, double eps, bool relu,
                               const c10::optional<at::Tensor>& run_mean,
                               const c10::optional<at::Tensor>& run_var,
                               double momentum,
                               const c10::optional<at::Tensor>& res) {
  check_bf16(x, "x"); check_bf16(gamma, "gamma"); check_bf16(beta, "beta");
  int C = x.size(-1);
  TORCH_CHECK(C % 8 == 0, "bn needs C % 8 == 0");
  i64 R = x.numel() / C;
  auto f32 = x.options().dtype(at::kFloat);
  auto part = cached_part(0, C, f32);        // bn_stats re-zeros it
  auto sum = part[0], sumsq = part[1];
  auto stats = at::empty({4, C}, f32);       // mean/rstd/scale/shift
  auto mean = stats[0], rstd = stats[1], scale = stats[2], shift = stats[3];
  auto y = at::empty_like(x);
  // reduce geometry: thread = 8-channel chunk; TC chunks x RS row-subs per
  // block; target ~2048 blocks total
  // A/B verdict: an 8-wide vectorized reduce (thread per 8-channel chunk)
  // measured WORSE at every geometry tried (70-220 us vs 48 us avg): the
  // scalar thread-per-channel reduce is already near bandwidth-bound
  // (coalesced 2 B/lane across a 256-lane channel window) and the wide
  // version only multiplied end-of-thread atomics. Kept scalar.
  int rpb = std::max(32, int(R / std::max<i64>(1, 2048 / ceil_div(C, 256))));
  auto s = cur_stream();
  dim3 rgrid(ceil_div(C, 256), ceil_div(R, (i64)rpb));
  hipLaunchKernelGGL(bn_reduce_kernel, rgrid, dim3(256), 0, s, bfp(x),
                     sum.data_ptr<float>(), sumsq.data_ptr<float>(), R, C,
                     rpb);
  float* rmp = run_mean ? run_mean->data_ptr<float>() : nullptr;
  float* rvp = run_var ? run_var->data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(bn_stats_kernel, dim3(ceil_div(C, 256)), dim3(256), 0, s,
                     sum.data_ptr<float>(), sumsq.data_ptr<float>(),
                     bfp(gamma), bfp(beta), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), scale.data_ptr<float>(),
                     shift.data_ptr<float>(), rmp, rvp,
                     R, C, float(eps), float(momentum));
  i64 total8 = x.numel() / 8;
  auto sc4 = reinterpret_cast<const float4*>(scale.data_ptr<float>());
  auto sh4 = reinterpret_cast<const float4*>(shift.data_ptr<float>());
  dim3 agrid(int((total8 + 255) / 256));
  const u16* rp = res ? bfp(*res) : nullptr;
  if (res && relu)
    hipLaunchKernelGGL((bn_apply_kernel<true, true>), agrid, dim3(256), 0, s,
                       bfp(x), rp, sc4, sh4, bfp_mut(y), total8, C);
  else if (res)
    hipLaunchKernelGGL((bn_apply_kernel<false, true>), agrid, dim3(256), 0,
                       s, bfp(x), rp, sc4, sh4, bfp_mut(y), total8, C);
  else if (relu)
    hipLaunchKernelGGL((bn_apply_kernel<true, false>), agrid, dim3(256), 0,
                       s, bfp(x), nullptr, sc4, sh4, bfp_mut(y), total8, C);
  else
    hipLaunchKernelGGL((bn_apply_kernel<false, false>), agrid, dim3(256), 0,
                       s, bfp(x), nullptr, sc4, sh4, bfp_mut(y), total8, C);
  return {y, mean, rstd};
}

// --------------------------------------------------------- BN bwd (K17)
// reduce dy and dy*xhat per channel -> dbeta/dgamma; then
// dx = rstd*gamma * (dy - mean(dy) - xhat * mean(dy*xhat))
template <bool RELU>
__global__ void bn_bwd_reduce_kernel(const u16* __restrict__ dy,
                                     const u16* __restrict__ x,
                                     const u16* __restrict__ yr,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     float* __restrict__ sdy,
                                     float* __restrict__ sdyx, i64 R, int C,
                                     int rows_per_block) {
  __shared__ float la[256], lb[256];
  int t = threadIdx.x;
  int W = 1, c = blockIdx.x * blockDim.x + t, w = 0;
  if (C < 256 && (C & (C - 1)) == 0) {      // see bn_reduce_kernel note
    W = 256 / C; c = t & (C - 1); w = t / C;
  } else if (c >= C) {
    return;
  }
  i64 r0 = (i64)blockIdx.y * rows_per_block;
  i64 r1 = min(R, r0 + rows_per_block);
  float a0 = 0.f, a1 = 0.f, b0 = 0.f, b1 = 0.f;
  float m = mean[c], rs = rstd[c];
  i64 r = r0 + 2 * w;
  for (; r + 2 <= r1; r += 2 * W) {
    float d0 = bf2f(dy[r * C + c]), d1 = bf2f(dy[(r + 1) * C + c]);
    if (RELU) {
      d0 = bf2f(yr[r * C + c]) > 0.f ? d0 : 0.f;
      d1 = bf2f(yr[(r + 1) * C + c]) > 0.f ? d1 : 0.f;
    }
    float h0 = (bf2f(x[r * C + c]) - m) * rs;
    float h1 = (bf2f(x[(r + 1) * C + c]) - m) * rs;
    a0 += d0; b0 += d0 * h0; a1 += d1; b1 += d1 * h1;
  }
  if (r < r1) {
    float d = bf2f(dy[r * C + c]);
    if (RELU) d = bf2f(yr[r * C + c]) > 0.f ? d : 0.f;
    float h = (bf2f(x[r * C + c]) - m) * rs;
    a0 += d; b0 += d * h;
  }
  float aa = a0 + a1, bb = b0 + b1;
  if (W > 1) {
    la[t] = aa; lb[t] = bb;
    __syncthreads();
    if (t >= C) return;
    for (int j = 1; j < W; ++j) { aa += la[t + j * C]; bb += lb[t + j * C]; }
  }
  int slot = blockIdx.y & 255;
  atomicAdd(&sdy[(i64)slot * C + c], aa);
  atomicAdd(&sdyx[(i64)slot * C + c], bb);
}

// One pass over the 256-way partials: emits dgamma/dbeta and the three
// per-channel dx coefficients (dx = A*dy_masked + B*x + D), replacing two
// slot-reduce launches + a param-grad launch + 40 scalar loads per 8
// elements in the old apply kernel.
__global__ void bn_bwd_stats_kernel(float* __restrict__ sdyp,
                                    float* __restrict__ sdyxp,
                                    const u16* __restrict__ gamma,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    float* __restrict__ cA,
                                    float* __restrict__ cB,
                                    float* __restrict__ cD,
                                    u16* __restrict__ dgamma,
                                    u16* __restrict__ dbeta, int C,
                                    float invR) {
  __shared__ float la[256], lb[256];
  int tid = threadIdx.x;
  int W = 1, c = blockIdx.x * blockDim.x + tid, w = 0;
  if (C < 256 && (C & (C - 1)) == 0) {    // see bn_reduce_kernel note
    W = 256 / C; c = tid & (C - 1); w = tid / C;
  } else if (c >= C) {
    return;
  }
  float a = 0.f, b = 0.f;
#pragma unroll 8
  for (int t = w; t < 256; t += W) {
    a += sdyp[(i64)t * C + c];
    b += sdyxp[(i64)t * C + c];
  }
#pragma unroll 8
  for (int t = w; t < 256; t += W) {  // cached_part contract: leave zeroed
    sdyp[(i64)t * C + c] = 0.f;
    sdyxp[(i64)t * C + c] = 0.f;
  }
  if (W > 1) {
    la[tid] = a; lb[tid] = b;
    __syncthreads();
    if (tid >= C) return;
    for (int j = 1; j < W; ++j) { a += la[tid + j * C]; b += lb[tid + j * C]; }
  }
  dgamma[c] = f2bf(b);
  dbeta[c] = f2bf(a);
  float g = bf2f(gamma[c]), rs = rstd[c], m = mean[c];
  float A = rs * g;
  float B = -rs * rs * g * b * invR;
  float D = -A * a * invR - m * B;
  cA[c] = A; cB[c] = B; cD[c] = D;
}

template <bool RELU>
__global__ void bn_bwd_apply_kernel(const u16* __restrict__ dy,
                                    const u16* __restrict__ x,
                                    const u16* __restrict__ yr,
                                    const float4* __restrict__ cA4,
                                    const float4* __restrict__ cB4,
                                    const float4* __restrict__ cD4,
                                    u16* __restrict__ dx, i64 total8, int C) {
  i64 id = (i64)blockIdx.x * blockDim.x + threadIdx.x;
  if (id >= total8) return;
  int nch = C / 8;
  int c8 = int(id % nch) * 8;
  i64 o = (id / nch) * C + c8;
  s16x8 dv = *reinterpret_cast<const s16x8*>(dy + o);
  s16x8 xv = *reinterpret_cast<const s16x8*>(x + o);
  s16x8 yv{};
  if (RELU) yv = *reinterpret_cast<const s16x8*>(yr + o);
  float4 aA = cA4[c8 / 4], aB = cA4[c8 / 4 + 1];
  float4 bA = cB4[c8 / 4], bB = cB4[c8 / 4 + 1];
  float4 dA = cD4[c8 / 4], dB = cD4[c8 / 4 + 1];
  float A[8] = {aA.x, aA.y, aA.z, aA.w, aB.x, aB.y, aB.z, aB.w};
  float B[8] = {bA.x, bA.y, bA.z, bA.w, bB.x, bB.y, bB.z, bB.w};
  float D[8] = {dA.x, dA.y, dA.z, dA.w, dB.x, dB.y, dB.z, dB.w};
  s16x8 out;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    float d = bf2f(u16(dv[j]));
    if (RELU) d = bf2f(u16(yv[j])) > 0.f ? d : 0.f;
    float v = A[j] * d + B[j] * bf2f(u16(xv[j])) + D[j];
    out[j] = short(f2bf(v));
  }
  *reinterpret_cast<s16x8*>(dx + o) = out;
}

std::vector<at::Tensor> bn_bwd(const at::Tensor& dy, const at::Tensor& x,
                               const at::Tensor& gamma, const at::Tensor& mean,
                               const at::Tensor& rstd, const at::Tensor& yr,
                               bool relu,
                               const c10::optional<at::Tensor>& dg_out,
                               const c10::optional<at::Tensor>& db_out) {
  check_bf16(dy, "dy"); check_bf16(x, "x");
  int C = dy.size(-1);
  TORCH_CHECK(C % 8 == 0, "bn needs C % 8 == 0");
  i64 R = dy.numel() / C;
  auto f32 = dy.options().dtype(at::kFloat);
  auto bpart = cached_part(1, C, f32);       // bn_bwd_stats re-zeros it
  auto sdyp = bpart[0], sdyxp = bpart[1];
  auto coef = at::empty({3, C}, f32);
  auto cA = coef[0], cB = coef[1], cD = coef[2];
  auto dx = at::empty_like(dy);
  auto dgamma = dg_out ? *dg_out : at::empty({C}, dy.options());
  auto dbeta = db_out ? *db_out : at::empty({C}, dy.options());
  int rpb = std::max(32, int(R / std::max<i64>(1, 2048 / ceil_div(C, 256))));
  auto s = cur_stream();
  dim3 rgrid(ceil_div(C, 256), ceil_div(R, (i64)rpb));
  const u16* yp = relu ? bfp(yr) : nullptr;
  if (relu)
    hipLaunchKernelGGL(bn_bwd_reduce_kernel<true>, rgrid, dim3(256), 0, s,
                       bfp(dy), bfp(x), yp, mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), sdyp.data_ptr<float>(),
                       sdyxp.data_ptr<float>(), R, C, rpb);
  else
    hipLaunchKernelGGL(bn_bwd_reduce_kernel<false>, rgrid, dim3(256), 0, s,
                       bfp(dy), bfp(x), yp, mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), sdyp.data_ptr<float>(),
                       sdyxp.data_ptr<float>(), R, C, rpb);
  hipLaunchKernelGGL(bn_bwd_stats_kernel, dim3(ceil_div(C, 256)), dim3(256),
                     0, s, sdyp.data_ptr<float>(), sdyxp.data_ptr<float>(),
                     bfp(gamma), mean.data_ptr<float>(),
                     rstd.data_ptr<float>(), cA.data_ptr<float>(),
                     cB.data_ptr<float>(), cD.data_ptr<float>(),
                     bfp_mut(dgamma), bfp_mut(dbeta), C,
                     float(1.0 / double(R)));
  i64 total8 = dy.numel() / 8;
  auto a4 = reinterpret_cast<const float4*>(cA.data_ptr<float>());
  auto b4 = reinterpret_cast<const float4*>(cB.data_ptr<float>());
  auto d4 = reinterpret_cast<const float4*>(cD.data_ptr<float>());
  if (relu)
    hipLaunchKernelGGL(bn_bwd_apply_kernel<true>,
                       dim3(int((total8 + 255) / 256)), dim3(256), 0, s,
                       bfp(dy), bfp(x), yp, a4, b4, d4, bfp_mut(dx), total8,
                       C);
  else
    hipLaunchKernelGGL(bn_bwd_apply_kernel<false>,
                       dim3(int((total8 + 255) / 256)), dim3(256), 0, s,
                       bfp(dy), bfp(x), yp, a4, b4, d4, bfp_mut(dx), total8,
                       C);
  return {dx, dgamma, dbeta};
}

}  // namespace dvgg
