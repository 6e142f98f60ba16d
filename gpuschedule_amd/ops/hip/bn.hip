// Fused BatchNorm2d (training) for gfx950, NHWC/channels_last layout.
//
// torch's native batch_norm kernels were 32% of the whole trace-replay
// bench (collect_statistics + transform_input + backward as separate
// unfused passes).  Here: channels are the INNERMOST dim
// (channels_last), so every 16-B vector load covers 8 consecutive
// channels; per-channel fp32 statistics accumulate in REGISTERS
// (each thread owns a fixed 8-channel group because the grid stride
// 2048 is a multiple of C for every power-of-2 C), merge once per
// block through LDS, and hit global memory with one atomicAdd per
// channel per block.  Apply kernels stage the per-channel scale/shift
// tables in LDS and fuse the ReLU (forward) / ReLU mask (backward).
//
// Memory passes: fwd = read x (stats) + read x/write y (apply);
// bwd = read dy,x (stats) + read dy,x / write dx (apply) — each pass
// at the vectorized-streaming roofline.
#include <torch/extension.h>
#include <c10/cuda/CUDAStream.h>
#include "common.h"

namespace {

constexpr int NT = 256;
constexpr int PASS = NT * 8;  // elements per block pass (2048)

// ---- per-channel sum/sumsq (or sum_dy/sum_dy_xhat) ----------------
// REGPATH requires PASS % C == 0 (all ResNet widths); BWD computes
// backward sums instead; RELU masks dy by (y > 0).
// RELU backward mask: read from y when given, else RECOMPUTED as
// (x*scale + shift) > 0 (valid only without a fused residual) — drops
// a full activation stream from the backward stats/apply reads.
template <bool BWD, bool RELU>
__global__ __launch_bounds__(NT) void bn_stats_kernel(
    const short* __restrict__ x, const short* __restrict__ dy,
    const short* __restrict__ y, const float* __restrict__ mean,
    const float* __restrict__ rstd, const float* __restrict__ gamma,
    const float* __restrict__ beta, float* __restrict__ out_a,
    float* __restrict__ out_b, long E, int C) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* lds_a = (float*)smem;          // [C]
  float* lds_b = lds_a + C;             // [C]
  for (int i = threadIdx.x; i < 2 * C; i += NT) lds_a[i] = 0.f;
  __syncthreads();

  const int t8 = threadIdx.x * 8;
  const int ch0 = t8 % C;               // fixed: PASS % C == 0
  float acc_a[8], acc_b[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) { acc_a[j] = 0.f; acc_b[j] = 0.f; }
  float mu[8], rs[8], msc[8], msh[8];
  if (BWD) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      mu[j] = mean[ch0 + j];
      rs[j] = rstd[ch0 + j];
    }
    if (RELU && !y) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        msc[j] = gamma[ch0 + j] * rs[j];
        msh[j] = beta[ch0 + j] - mu[j] * msc[j];
      }
    }
  }

  long base = (long)blockIdx.x * PASS + t8;
  long stride = (long)gridDim.x * PASS;
  for (long e = base; e + 7 < E; e += stride) {
    short8 xv = *reinterpret_cast<const short8*>(x + e);
    if (!BWD) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf16_to_f32(xv[j]);
        acc_a[j] += f;
        acc_b[j] += f * f;
      }
    } else {
      short8 dv = *reinterpret_cast<const short8*>(dy + e);
      short8 yv;
      if (RELU && y) yv = *reinterpret_cast<const short8*>(y + e);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float d = bf16_to_f32(dv[j]);
        float xf = bf16_to_f32(xv[j]);
        if (RELU) {
          bool dead = y ? (bf16_to_f32(yv[j]) <= 0.f)
                        : (xf * msc[j] + msh[j] <= 0.f);
          if (dead) d = 0.f;
        }
        float xh = (xf - mu[j]) * rs[j];
        acc_a[j] += d;
        acc_b[j] += d * xh;
      }
    }
  }
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    atomicAdd(lds_a + ch0 + j, acc_a[j]);
    atomicAdd(lds_b + ch0 + j, acc_b[j]);
  }
  __syncthreads();
  // per-block partial slot [blockIdx][2C]: no pre-zeroed global
  // accumulator, no global atomics; finalize sums the slots.
  float* slot = out_a + (long)blockIdx.x * 2 * C;
  for (int i = threadIdx.x; i < 2 * C; i += NT) slot[i] = lds_a[i];
}

// ---- apply: y = x * scale[c] + shift[c] (+relu) -------------------
template <bool RELU, bool RES>
__global__ __launch_bounds__(NT) void bn_apply_kernel(
    const short* __restrict__ x, const short* __restrict__ res,
    short* __restrict__ y,
    const float* __restrict__ scale, const float* __restrict__ shift,
    long E, int C) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* s_sc = (float*)smem;
  float* s_sh = s_sc + C;
  for (int i = threadIdx.x; i < C; i += NT) {
    s_sc[i] = scale[i];
    s_sh[i] = shift[i];
  }
  __syncthreads();
  const int t8 = threadIdx.x * 8;
  const int ch0 = t8 % C;
  long base = (long)blockIdx.x * PASS + t8;
  long stride = (long)gridDim.x * PASS;
  for (long e = base; e + 7 < E; e += stride) {
    short8 xv = *reinterpret_cast<const short8*>(x + e);
    short8 rv;
    if (RES) rv = *reinterpret_cast<const short8*>(res + e);
    short8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf16_to_f32(xv[j]) * s_sc[ch0 + j] + s_sh[ch0 + j];
      if (RES) f += bf16_to_f32(rv[j]);
      if (RELU) f = fmaxf(f, 0.f);
      out[j] = f32_to_bf16(f);
    }
    *reinterpret_cast<short8*>(y + e) = out;
  }
}

// ---- backward apply:
// dx = g*rstd * (dy' - sum_dy/Nc - xhat * sum_dyxhat/Nc)
template <bool RELU, bool RES>
__global__ __launch_bounds__(NT) void bn_bwd_apply_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    const short* __restrict__ y, short* __restrict__ dx,
    short* __restrict__ dres,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    const float* __restrict__ sum_dy,
    const float* __restrict__ sum_dyxh, long E, int C, float inv_cnt) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  // per-channel constants: k1 = g*rstd, c1 = sum_dy/Nc,
  // c2 = sum_dyxh/Nc, mu, rs (+ shift when the relu mask is
  // recomputed from x instead of read from y)
  float* s_k1 = (float*)smem;
  float* s_c1 = s_k1 + C;
  float* s_c2 = s_c1 + C;
  float* s_mu = s_c2 + C;
  float* s_rs = s_mu + C;
  float* s_sh = s_rs + C;
  for (int i = threadIdx.x; i < C; i += NT) {
    float k1 = gamma[i] * rstd[i];
    s_k1[i] = k1;
    s_c1[i] = sum_dy[i] * inv_cnt;
    s_c2[i] = sum_dyxh[i] * inv_cnt;
    s_mu[i] = mean[i];
    s_rs[i] = rstd[i];
    if (RELU && beta) s_sh[i] = beta[i] - mean[i] * k1;
  }
  __syncthreads();
  const int t8 = threadIdx.x * 8;
  const int ch0 = t8 % C;
  long base = (long)blockIdx.x * PASS + t8;
  long stride = (long)gridDim.x * PASS;
  for (long e = base; e + 7 < E; e += stride) {
    short8 dv = *reinterpret_cast<const short8*>(dy + e);
    short8 xv = *reinterpret_cast<const short8*>(x + e);
    short8 yv;
    if (RELU && y) yv = *reinterpret_cast<const short8*>(y + e);
    short8 out, rout;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int c = ch0 + j;
      float d = bf16_to_f32(dv[j]);
      if (RELU) {
        bool dead = y ? (bf16_to_f32(yv[j]) <= 0.f)
                      : (bf16_to_f32(xv[j]) * s_k1[c] + s_sh[c] <= 0.f);
        if (dead) d = 0.f;
      }
      if (RES) rout[j] = f32_to_bf16(d);
      float xh = (bf16_to_f32(xv[j]) - s_mu[c]) * s_rs[c];
      out[j] = f32_to_bf16(s_k1[c] * (d - s_c1[c] - xh * s_c2[c]));
    }
    *reinterpret_cast<short8*>(dx + e) = out;
    if (RES) *reinterpret_cast<short8*>(dres + e) = rout;
  }
}

// finalize: sums -> mean/rstd/scale/shift (+running-stat update), one
// kernel instead of ~10 C-length torch ops per BN layer per step.
// 64 channels x 4 slot-lanes per block: the slot loop is 4-way
// parallel per channel (a serial P-loop with few threads was the
// LN-backward mistake; rule: reductions get chip-filling parallelism).
__global__ void bn_finalize_kernel(
    const float* __restrict__ part, int P,
    const float* __restrict__ gamma,
    const float* __restrict__ beta, float* __restrict__ mean_out,
    float* __restrict__ rstd_out, float* __restrict__ scale,
    float* __restrict__ shift, float* __restrict__ running_mean,
    float* __restrict__ running_var, int C, float inv_cnt,
    float unbias, float momentum, float eps) {
  __shared__ float sh[2][64][16];
  const int cl = threadIdx.x & 63, pl = threadIdx.x >> 6;
  const int c = blockIdx.x * 64 + cl;
  float s1 = 0.f, s2 = 0.f;
  if (c < C)
    for (int p = pl; p < P; p += 16) {
      s1 += part[(long)p * 2 * C + c];
      s2 += part[(long)p * 2 * C + C + c];
    }
  sh[0][cl][pl] = s1;
  sh[1][cl][pl] = s2;
  __syncthreads();
  if (pl != 0 || c >= C) return;
  s1 = 0.f; s2 = 0.f;
#pragma unroll
  for (int j = 0; j < 16; ++j) { s1 += sh[0][cl][j]; s2 += sh[1][cl][j]; }
  float mu = s1 * inv_cnt;
  float var = s2 * inv_cnt - mu * mu;
  float rs = rsqrtf(var + eps);
  mean_out[c] = mu;
  rstd_out[c] = rs;
  float sc = gamma[c] * rs;
  scale[c] = sc;
  shift[c] = beta[c] - mu * sc;
  if (running_mean) {
    running_mean[c] = (1.f - momentum) * running_mean[c] + momentum * mu;
    running_var[c] = (1.f - momentum) * running_var[c]
        + momentum * var * unbias;
  }
}

__global__ void bn_reduce_slots_kernel(
    const float* __restrict__ part, int P, float* __restrict__ out,
    int n) {
  __shared__ float sh[64][16];
  const int cl = threadIdx.x & 63, pl = threadIdx.x >> 6;
  const int i = blockIdx.x * 64 + cl;
  float s = 0.f;
  if (i < n)
    for (int p = pl; p < P; p += 16) s += part[(long)p * n + i];
  sh[cl][pl] = s;
  __syncthreads();
  if (pl != 0 || i >= n) return;
  s = 0.f;
#pragma unroll
  for (int j = 0; j < 16; ++j) s += sh[cl][j];
  out[i] = s;
}

__global__ void bn_scale_shift_kernel(
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const float* __restrict__ gamma, const float* __restrict__ beta,
    float* __restrict__ scale, float* __restrict__ shift, int C) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float sc = gamma[c] * rstd[c];
  scale[c] = sc;
  shift[c] = beta[c] - mean[c] * sc;
}

int grid_for(long E) {
  // enough blocks to saturate HBM, few enough that the finalize
  // slot-reduce stays cheap
  long b = (E + PASS - 1) / PASS;
  return (int)std::min<long>(b > 0 ? b : 1, 304);
}

void check_bn(const torch::Tensor& x, int C) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(C % 8 == 0 && PASS % C == 0,
              "fused BN needs C%8==0 and 2048%C==0, got C=", C);
}

}  // namespace

// x viewed as [E/C rows][C]; returns (y, mean, rstd) in training mode
// or (y,) with given stats.  gamma/beta/mean/rstd are fp32 [C].
std::vector<torch::Tensor> bn_fwd(torch::Tensor x, torch::Tensor gamma,
                                  torch::Tensor beta,
                                  c10::optional<torch::Tensor> use_mean,
                                  c10::optional<torch::Tensor> use_rstd,
                                  c10::optional<torch::Tensor> run_mean,
                                  c10::optional<torch::Tensor> run_var,
                                  c10::optional<torch::Tensor> residual,
                                  double momentum, double eps, bool relu) {
  const int C = x.size(-1);
  check_bn(x, C);
  const long E = x.numel();
  auto fopt = x.options().dtype(torch::kFloat32);
  auto stream = c10::cuda::getCurrentCUDAStream();
  const int cgrid = (C + NT - 1) / NT;
  torch::Tensor mean, rstd;
  auto scale = torch::empty({C}, fopt);
  auto shift = torch::empty({C}, fopt);
  if (use_mean.has_value()) {
    mean = *use_mean;
    rstd = *use_rstd;
    hipLaunchKernelGGL(bn_scale_shift_kernel, dim3(cgrid), dim3(NT), 0,
                       stream.stream(), mean.data_ptr<float>(),
                       rstd.data_ptr<float>(), gamma.data_ptr<float>(),
                       beta.data_ptr<float>(), scale.data_ptr<float>(),
                       shift.data_ptr<float>(), C);
  } else {
    int sgrid = grid_for(E);
    auto part = torch::empty({sgrid, 2 * C}, fopt);
    hipLaunchKernelGGL((bn_stats_kernel<false, false>),
                       dim3(sgrid), dim3(NT), 2 * C * 4,
                       stream.stream(), (const short*)x.data_ptr(),
                       nullptr, nullptr, nullptr, nullptr, nullptr,
                       nullptr, part.data_ptr<float>(), nullptr, E, C);
    long cnt = E / C;
    mean = torch::empty({C}, fopt);
    rstd = torch::empty({C}, fopt);
    float* rm = run_mean.has_value() ? run_mean->data_ptr<float>()
                                     : nullptr;
    float* rv = run_var.has_value() ? run_var->data_ptr<float>()
                                    : nullptr;
    hipLaunchKernelGGL(bn_finalize_kernel, dim3((C + 63) / 64),
                       dim3(1024), 0,
                       stream.stream(), part.data_ptr<float>(), sgrid,
                       gamma.data_ptr<float>(), beta.data_ptr<float>(),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       rm, rv, C, 1.0f / (float)cnt,
                       (float)cnt / (float)std::max<long>(cnt - 1, 1),
                       (float)momentum, (float)eps);
  }
  auto y = torch::empty_like(x);
  const short* res_p = residual.has_value()
      ? (const short*)residual->data_ptr() : nullptr;
#define BN_APPLY(RL, RS)                                                 \
  hipLaunchKernelGGL((bn_apply_kernel<RL, RS>), dim3(grid_for(E)),       \
                     dim3(NT), 2 * C * 4, stream.stream(),               \
                     (const short*)x.data_ptr(), res_p,                  \
                     (short*)y.data_ptr(), scale.data_ptr<float>(),      \
                     shift.data_ptr<float>(), E, C)
  if (relu) { if (res_p) BN_APPLY(true, true); else BN_APPLY(true, false); }
  else { if (res_p) BN_APPLY(false, true); else BN_APPLY(false, false); }
#undef BN_APPLY
  HIP_CHECK_LAST();
  return {y, mean, rstd};
}

// returns (dx, sum_dy (=dbeta), sum_dyxh (=dgamma)); y needed iff relu
// relu backward needs the forward mask: pass y, OR (without a fused
// residual) pass beta instead and the kernels recompute the mask from
// x — one fewer full activation stream read in both backward passes.
std::vector<torch::Tensor> bn_bwd(torch::Tensor dy, torch::Tensor x,
                                  c10::optional<torch::Tensor> y,
                                  torch::Tensor mean, torch::Tensor rstd,
                                  torch::Tensor gamma, bool relu,
                                  bool want_dres,
                                  c10::optional<torch::Tensor> beta) {
  const int C = x.size(-1);
  const float* beta_p = nullptr;
  if (beta.has_value() && beta->defined())
    beta_p = beta->data_ptr<float>();
  const short* y_mask = nullptr;  // forward activations for the mask
  if (relu) {
    TORCH_CHECK(y.has_value() || beta_p,
                "bn_bwd: relu needs y or beta (mask recompute)");
    if (y.has_value()) y_mask = (const short*)y->data_ptr();
  }
  check_bn(x, C);
  const long E = x.numel();
  auto fopt = x.options().dtype(torch::kFloat32);
  auto stream = c10::cuda::getCurrentCUDAStream();
  int sgrid = grid_for(E);
  auto part = torch::empty({sgrid, 2 * C}, fopt);
  auto sums = torch::empty({2 * C}, fopt);
  auto sum_dy = sums.narrow(0, 0, C);
  auto sum_dyxh = sums.narrow(0, C, C);
  const short* y_p = y_mask;
  if (relu) {
    hipLaunchKernelGGL((bn_stats_kernel<true, true>), dim3(sgrid),
                       dim3(NT), 2 * C * 4, stream.stream(),
                       (const short*)x.data_ptr(),
                       (const short*)dy.data_ptr(), y_p,
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       gamma.data_ptr<float>(), beta_p,
                       part.data_ptr<float>(), nullptr, E, C);
  } else {
    hipLaunchKernelGGL((bn_stats_kernel<true, false>), dim3(sgrid),
                       dim3(NT), 2 * C * 4, stream.stream(),
                       (const short*)x.data_ptr(),
                       (const short*)dy.data_ptr(), nullptr,
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       nullptr, nullptr,
                       part.data_ptr<float>(), nullptr, E, C);
  }
  hipLaunchKernelGGL(bn_reduce_slots_kernel,
                     dim3((2 * C + 63) / 64), dim3(1024), 0,
                     stream.stream(), part.data_ptr<float>(), sgrid,
                     sums.data_ptr<float>(), 2 * C);
  auto dx = torch::empty_like(x);
  torch::Tensor dres;
  short* dres_p = nullptr;
  if (want_dres) {
    dres = torch::empty_like(x);
    dres_p = (short*)dres.data_ptr();
  }
  float inv_cnt = 1.0f / (float)(E / C);
#define BN_BWD_APPLY(RL, RS, YP)                                         \
  hipLaunchKernelGGL((bn_bwd_apply_kernel<RL, RS>), dim3(grid_for(E)),   \
                     dim3(NT), 6 * C * 4, stream.stream(),               \
                     (const short*)dy.data_ptr(),                        \
                     (const short*)x.data_ptr(), YP,                     \
                     (short*)dx.data_ptr(), dres_p,                      \
                     mean.data_ptr<float>(),                             \
                     rstd.data_ptr<float>(), gamma.data_ptr<float>(),    \
                     beta_p, sum_dy.data_ptr<float>(),                   \
                     sum_dyxh.data_ptr<float>(), E, C, inv_cnt)
  if (relu) {
    if (want_dres) BN_BWD_APPLY(true, true, y_p);
    else BN_BWD_APPLY(true, false, y_p);
  } else {
    if (want_dres) BN_BWD_APPLY(false, true, nullptr);
    else BN_BWD_APPLY(false, false, nullptr);
  }
#undef BN_BWD_APPLY
  HIP_CHECK_LAST();
  if (want_dres) return {dx, sum_dy, sum_dyxh, dres};
  return {dx, sum_dy, sum_dyxh};
}
