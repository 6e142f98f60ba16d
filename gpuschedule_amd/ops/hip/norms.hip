// Fused LayerNorm / RMSNorm forward+backward for gfx950 (MI355X).
//
// Memory-bound kernels: the design goal is the HBM roofline, so all
// bf16 traffic is vectorized as short8 (16 B/lane) per guideline G13,
// rows are block-resident (one 256-thread block per row, chunks held
// in registers with compile-time indexing), and reductions are 64-wide
// wave shuffles + one LDS hop (no atomics in the row path).
// Grad wrt gamma/beta uses per-block partials reduced by a second tiny
// kernel (guideline 12: no global atomic contention).
#include <torch/extension.h>
#include <c10/cuda/CUDAStream.h>
#include "common.h"

namespace {

constexpr int NT = 256;    // threads per block (4 waves)
constexpr int VEC = 8;     // bf16 per vector load
constexpr int CHUNK = NT * VEC;  // elements covered per chunk pass

// ---------------- forward ------------------------------------------
// one block per row; NCHUNK = ceil(D / 2048) template-dispatched so the
// row stays in registers (runtime-indexed arrays go to scratch).
template <int NCHUNK, bool RMS, bool RES>
__global__ __launch_bounds__(NT) void norm_fwd_kernel(
    const short* __restrict__ x, const short* __restrict__ res,
    short* __restrict__ sum_out, const short* __restrict__ gamma,
    const short* __restrict__ beta, short* __restrict__ y,
    float* __restrict__ mean_out, float* __restrict__ rstd_out,
    int n_rows, int D, float eps) {
  __shared__ float scratch[16];
  for (int row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const short* xr = x + (long)row * D;
    short8 xv[NCHUNK];
    float sum = 0.f, sq = 0.f;
#pragma unroll
    for (int c = 0; c < NCHUNK; ++c) {
      int i = c * CHUNK + threadIdx.x * VEC;
      if (i < D) {
        xv[c] = *reinterpret_cast<const short8*>(xr + i);
        if (RES) {
          // fused residual add; the summed activation IS the tensor
          // backward normalizes against, so write it out
          short8 rv = *reinterpret_cast<const short8*>(
              res + (long)row * D + i);
#pragma unroll
          for (int j = 0; j < VEC; ++j)
            xv[c][j] = f32_to_bf16(bf16_to_f32(xv[c][j]) +
                                   bf16_to_f32(rv[j]));
          *reinterpret_cast<short8*>(sum_out + (long)row * D + i) =
              xv[c];
        }
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float f = bf16_to_f32(xv[c][j]);
          sum += f;
          sq += f * f;
        }
      } else {
#pragma unroll
        for (int j = 0; j < VEC; ++j) xv[c][j] = 0;
      }
    }
    float mean = 0.f;
    if (!RMS) {
      mean = block_reduce_sum<NT / WAVE>(sum, scratch) / D;
      __syncthreads();
    }
    float ms = block_reduce_sum<NT / WAVE>(sq, scratch) / D;
    float rstd = rsqrtf((RMS ? ms : ms - mean * mean) + eps);
    if (threadIdx.x == 0) {
      if (!RMS && mean_out) mean_out[row] = mean;
      if (rstd_out) rstd_out[row] = rstd;
    }
    short* yr = y + (long)row * D;
#pragma unroll
    for (int c = 0; c < NCHUNK; ++c) {
      int i = c * CHUNK + threadIdx.x * VEC;
      if (i < D) {
        short8 gv = *reinterpret_cast<const short8*>(gamma + i);
        short8 out;
        if (!RMS) {
          short8 bv = *reinterpret_cast<const short8*>(beta + i);
#pragma unroll
          for (int j = 0; j < VEC; ++j) {
            float f = (bf16_to_f32(xv[c][j]) - mean) * rstd;
            out[j] = f32_to_bf16(f * bf16_to_f32(gv[j]) +
                                 bf16_to_f32(bv[j]));
          }
        } else {
#pragma unroll
          for (int j = 0; j < VEC; ++j) {
            float f = bf16_to_f32(xv[c][j]) * rstd;
            out[j] = f32_to_bf16(f * bf16_to_f32(gv[j]));
          }
        }
        *reinterpret_cast<short8*>(yr + i) = out;
      }
    }
    __syncthreads();
  }
}

// ---- small-D variants: one WAVE per row (D <= 1024) ---------------
// The block-per-row kernel leaves (1 - D/2048) of each block idle at
// small D (62% at BERT's 768).  Here: 4 rows per 256-thread block,
// pure wave-shuffle reductions, no LDS in the row path.
template <bool RMS, bool RES>
__global__ __launch_bounds__(NT) void norm_fwd_wave_kernel(
    const short* __restrict__ x, const short* __restrict__ res,
    short* __restrict__ sum_out, const short* __restrict__ gamma,
    const short* __restrict__ beta, short* __restrict__ y,
    float* __restrict__ mean_out, float* __restrict__ rstd_out,
    long n_rows, int D, float eps) {
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  for (long row = (long)blockIdx.x * 4 + wid; row < n_rows;
       row += (long)gridDim.x * 4) {
    const short* xr = x + row * D;
    short8 xv[2];
    float sum = 0.f, sq = 0.f;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int i = c * 512 + lane * VEC;
      if (i < D) {
        xv[c] = *reinterpret_cast<const short8*>(xr + i);
        if (RES) {
          short8 rv = *reinterpret_cast<const short8*>(
              res + row * D + i);
#pragma unroll
          for (int j = 0; j < VEC; ++j)
            xv[c][j] = f32_to_bf16(bf16_to_f32(xv[c][j]) +
                                   bf16_to_f32(rv[j]));
          *reinterpret_cast<short8*>(sum_out + row * D + i) = xv[c];
        }
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float f = bf16_to_f32(xv[c][j]);
          sum += f;
          sq += f * f;
        }
      }
    }
    float mean = 0.f;
    if (!RMS) mean = wave_reduce_sum(sum) / D;
    float ms = wave_reduce_sum(sq) / D;
    float rstd = rsqrtf((RMS ? ms : ms - mean * mean) + eps);
    if (lane == 0) {
      if (!RMS && mean_out) mean_out[row] = mean;
      if (rstd_out) rstd_out[row] = rstd;
    }
    short* yr = y + row * D;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int i = c * 512 + lane * VEC;
      if (i < D) {
        short8 gv = *reinterpret_cast<const short8*>(gamma + i);
        short8 out;
        if (!RMS) {
          short8 bv = *reinterpret_cast<const short8*>(beta + i);
#pragma unroll
          for (int j = 0; j < VEC; ++j)
            out[j] = f32_to_bf16(
                (bf16_to_f32(xv[c][j]) - mean) * rstd *
                    bf16_to_f32(gv[j]) + bf16_to_f32(bv[j]));
        } else {
#pragma unroll
          for (int j = 0; j < VEC; ++j)
            out[j] = f32_to_bf16(bf16_to_f32(xv[c][j]) * rstd *
                                 bf16_to_f32(gv[j]));
        }
        *reinterpret_cast<short8*>(yr + i) = out;
      }
    }
  }
}

template <bool RMS>
__global__ __launch_bounds__(NT) void norm_bwd_wave_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const short* __restrict__ gamma, const short* __restrict__ dadd,
    short* __restrict__ dx,
    float* __restrict__ part_dg, float* __restrict__ part_db,
    long pstride, long n_rows, int D) {
  // per-thread dgamma/dbeta accumulators for its fixed columns; the
  // block's 4 waves cover the SAME columns -> merge through LDS once.
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* merge = (float*)smem;  // [2][D]
  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  float acc_dg[2][VEC], acc_db[2][VEC];
#pragma unroll
  for (int c = 0; c < 2; ++c)
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      acc_dg[c][j] = 0.f;
      acc_db[c][j] = 0.f;
    }
  for (long row = (long)blockIdx.x * 4 + wid; row < n_rows;
       row += (long)gridDim.x * 4) {
    const short* dyr = dy + row * D;
    const short* xr = x + row * D;
    const float mu = RMS ? 0.f : mean[row];
    const float rs = rstd[row];
    short8 dyv[2], xv[2], gv[2];
    float c1 = 0.f, c2 = 0.f;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int i = c * 512 + lane * VEC;
      if (i < D) {
        dyv[c] = *reinterpret_cast<const short8*>(dyr + i);
        xv[c] = *reinterpret_cast<const short8*>(xr + i);
        gv[c] = *reinterpret_cast<const short8*>(gamma + i);
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float h = bf16_to_f32(dyv[c][j]) * bf16_to_f32(gv[c][j]);
          float xn = (bf16_to_f32(xv[c][j]) - mu) * rs;
          c1 += h;
          c2 += h * xn;
        }
      }
    }
    if (!RMS) c1 = wave_reduce_sum(c1) / D;
    c2 = wave_reduce_sum(c2) / D;
    short* dxr = dx + row * D;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      int i = c * 512 + lane * VEC;
      if (i < D) {
        short8 out;
        short8 dav;
        if (dadd)  // fused downstream residual-stream grad add
          dav = *reinterpret_cast<const short8*>(dadd + row * D + i);
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float d = bf16_to_f32(dyv[c][j]);
          float h = d * bf16_to_f32(gv[c][j]);
          float xn = (bf16_to_f32(xv[c][j]) - mu) * rs;
          float v = RMS ? rs * (h - xn * c2) : rs * (h - c1 - xn * c2);
          if (dadd) v += bf16_to_f32(dav[j]);
          out[j] = f32_to_bf16(v);
          acc_dg[c][j] += d * xn;
          acc_db[c][j] += d;
        }
        *reinterpret_cast<short8*>(dxr + i) = out;
      }
    }
  }
  // merge the 4 waves' column accumulators -> block partial slot
  for (int i = threadIdx.x; i < 2 * D; i += NT) merge[i] = 0.f;
  __syncthreads();
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    int i = c * 512 + lane * VEC;
    if (i < D) {
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        atomicAdd(merge + i + j, acc_dg[c][j]);
        atomicAdd(merge + D + i + j, acc_db[c][j]);
      }
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < D; i += NT) {
    part_dg[(long)blockIdx.x * pstride + i] = merge[i];
    if (part_db) part_db[(long)blockIdx.x * pstride + i] = merge[D + i];
  }
}

// ---------------- backward (dx + per-block dgamma/dbeta partials) ---
template <int NCHUNK, bool RMS>
__global__ __launch_bounds__(NT) void norm_bwd_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    const short* __restrict__ gamma, const short* __restrict__ dadd,
    short* __restrict__ dx,
    float* __restrict__ part_dg, float* __restrict__ part_db,
    long pstride, int n_rows, int D) {
  __shared__ float scratch[16];
  float acc_dg[NCHUNK][VEC];
  float acc_db[NCHUNK][VEC];
#pragma unroll
  for (int c = 0; c < NCHUNK; ++c)
#pragma unroll
    for (int j = 0; j < VEC; ++j) { acc_dg[c][j] = 0.f; acc_db[c][j] = 0.f; }

  for (int row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const short* dyr = dy + (long)row * D;
    const short* xr = x + (long)row * D;
    const float mu = RMS ? 0.f : mean[row];
    const float rs = rstd[row];
    short8 dyv[NCHUNK], xv[NCHUNK], gv[NCHUNK];
    float c1 = 0.f, c2 = 0.f;
#pragma unroll
    for (int c = 0; c < NCHUNK; ++c) {
      int i = c * CHUNK + threadIdx.x * VEC;
      if (i < D) {
        dyv[c] = *reinterpret_cast<const short8*>(dyr + i);
        xv[c] = *reinterpret_cast<const short8*>(xr + i);
        gv[c] = *reinterpret_cast<const short8*>(gamma + i);
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float h = bf16_to_f32(dyv[c][j]) * bf16_to_f32(gv[c][j]);
          float xn = (bf16_to_f32(xv[c][j]) - mu) * rs;
          c1 += h;
          c2 += h * xn;
        }
      }
    }
    if (!RMS) {
      c1 = block_reduce_sum<NT / WAVE>(c1, scratch) / D;
      __syncthreads();
    }
    c2 = block_reduce_sum<NT / WAVE>(c2, scratch) / D;
    short* dxr = dx + (long)row * D;
#pragma unroll
    for (int c = 0; c < NCHUNK; ++c) {
      int i = c * CHUNK + threadIdx.x * VEC;
      if (i < D) {
        short8 out;
        short8 dav;
        if (dadd)  // fused downstream residual-stream grad add
          dav = *reinterpret_cast<const short8*>(
              dadd + (long)row * D + i);
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          float d = bf16_to_f32(dyv[c][j]);
          float h = d * bf16_to_f32(gv[c][j]);
          float xn = (bf16_to_f32(xv[c][j]) - mu) * rs;
          float v = RMS ? rs * (h - xn * c2)
                        : rs * (h - c1 - xn * c2);
          if (dadd) v += bf16_to_f32(dav[j]);
          out[j] = f32_to_bf16(v);
          acc_dg[c][j] += d * xn;
          acc_db[c][j] += d;
        }
        *reinterpret_cast<short8*>(dxr + i) = out;
      }
    }
    __syncthreads();
  }
  // per-block partials (block owns slice blockIdx.x)
#pragma unroll
  for (int c = 0; c < NCHUNK; ++c) {
    int i = c * CHUNK + threadIdx.x * VEC;
    if (i < D) {
      float* pg = part_dg + (long)blockIdx.x * pstride + i;
#pragma unroll
      for (int j = 0; j < VEC; ++j) pg[j] = acc_dg[c][j];
      if (part_db) {
        float* pb = part_db + (long)blockIdx.x * pstride + i;
#pragma unroll
        for (int j = 0; j < VEC; ++j) pb[j] = acc_db[c][j];
      }
    }
  }
}

// partials [P][D2] f32 -> bf16 [D2] in ONE launch (vs the original
// zero-fill + device-atomic accumulate + cast trio): deterministic
// plain reads, no fp32 fill kernel per call.  D2 = 2*D for layernorm
// (dgamma and dbeta packed side by side in one buffer so both grads
// come out of a single reduction).  64 columns x 16 row-lanes per
// 1024-thread block keep the P loop 16-way parallel per column — a
// serial-P version with D2/256 blocks left most CUs idle (same lesson
// as the BN slot reduce).
__global__ void sum_cast_bf16_kernel(const float* __restrict__ part,
                                     short* __restrict__ out,
                                     int P, int D2) {
  __shared__ float sh[64][16];
  const int cl = threadIdx.x & 63, pl = threadIdx.x >> 6;
  const int i = blockIdx.x * 64 + cl;
  float s = 0.f;
  if (i < D2)
    for (int p = pl; p < P; p += 16) s += part[(long)p * D2 + i];
  sh[cl][pl] = s;
  __syncthreads();
  if (pl != 0 || i >= D2) return;
  s = 0.f;
#pragma unroll
  for (int j = 0; j < 16; ++j) s += sh[cl][j];
  out[i] = f32_to_bf16(s);
}

int pick_nchunk(int D) {
  int need = (D + CHUNK - 1) / CHUNK;
  for (int c : {1, 2, 4, 8})
    if (need <= c) return c;
  TORCH_CHECK(false, "norm: D too large (max ", 8 * CHUNK, "), got ", D);
  return -1;
}

#define DISPATCH_NCHUNK(NC, RMS, ...)                 \
  switch (NC) {                                       \
    case 1: { constexpr int kNC = 1; __VA_ARGS__; break; } \
    case 2: { constexpr int kNC = 2; __VA_ARGS__; break; } \
    case 4: { constexpr int kNC = 4; __VA_ARGS__; break; } \
    case 8: { constexpr int kNC = 8; __VA_ARGS__; break; } \
  }

void check_norm_inputs(const torch::Tensor& x, const torch::Tensor& gamma) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous(), "x must be contiguous CUDA");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16, "x must be bf16");
  TORCH_CHECK(gamma.scalar_type() == torch::kBFloat16, "gamma must be bf16");
  TORCH_CHECK(x.size(-1) % VEC == 0, "D must be a multiple of 8");
}

}  // namespace

// (y, sum_or_x, mean, rstd) = norm_fwd(x, residual?, gamma, beta?, ...)
// With a residual the normalized tensor is x+residual and is written
// out (backward normalizes against it; callers may reuse it as the
// residual stream).  RMS: mean slot returns rstd.
std::vector<torch::Tensor> norm_fwd(torch::Tensor x,
                                    c10::optional<torch::Tensor> residual,
                                    torch::Tensor gamma,
                                    c10::optional<torch::Tensor> beta,
                                    double eps, bool rms) {
  check_norm_inputs(x, gamma);
  const int D = x.size(-1);
  const long n_rows = x.numel() / D;
  auto y = torch::empty_like(x);
  auto fopt = x.options().dtype(torch::kFloat32);
  auto mean = rms ? torch::Tensor() : torch::empty({n_rows}, fopt);
  auto rstd = torch::empty({n_rows}, fopt);
  int nc = pick_nchunk(D);
  int grid = (int)std::min<long>(n_rows, 8192);
  auto stream = c10::cuda::getCurrentCUDAStream();
  const short* beta_p = nullptr;
  if (!rms) {
    TORCH_CHECK(beta.has_value(), "layernorm needs beta");
    beta_p = (const short*)beta->data_ptr();
  }
  const bool has_res = residual.has_value() && residual->defined();
  torch::Tensor sum_out;
  const short* res_p = nullptr;
  short* sum_p = nullptr;
  if (has_res) {
    TORCH_CHECK(residual->is_contiguous() &&
                residual->scalar_type() == torch::kBFloat16);
    sum_out = torch::empty_like(x);
    res_p = (const short*)residual->data_ptr();
    sum_p = (short*)sum_out.data_ptr();
  }
  if (D <= 1024) {  // wave-per-row fast path
    long wgrid = std::min<long>((n_rows + 3) / 4, 8192);
#define NORM_FWDW_LAUNCH(RMSB, RESB)                                      \
    hipLaunchKernelGGL((norm_fwd_wave_kernel<RMSB, RESB>),               \
                       dim3((int)wgrid), dim3(NT), 0, stream.stream(),   \
                       (const short*)x.data_ptr(), res_p, sum_p,         \
                       (const short*)gamma.data_ptr(), beta_p,           \
                       (short*)y.data_ptr(),                             \
                       RMSB ? nullptr : mean.data_ptr<float>(),          \
                       rstd.data_ptr<float>(), n_rows, D, (float)eps)
    if (rms) {
      if (has_res) NORM_FWDW_LAUNCH(true, true);
      else NORM_FWDW_LAUNCH(true, false);
    } else {
      if (has_res) NORM_FWDW_LAUNCH(false, true);
      else NORM_FWDW_LAUNCH(false, false);
    }
#undef NORM_FWDW_LAUNCH
    HIP_CHECK_LAST();
    return {y, has_res ? sum_out : x, mean.defined() ? mean : rstd,
            rstd};
  }
#define NORM_FWD_LAUNCH(kNC, RMSB, RESB)                                  \
  hipLaunchKernelGGL((norm_fwd_kernel<kNC, RMSB, RESB>), dim3(grid),      \
                     dim3(NT), 0, stream.stream(),                        \
                     (const short*)x.data_ptr(), res_p, sum_p,            \
                     (const short*)gamma.data_ptr(), beta_p,              \
                     (short*)y.data_ptr(),                                \
                     RMSB ? nullptr : mean.data_ptr<float>(),             \
                     rstd.data_ptr<float>(), (int)n_rows, D, (float)eps)
  DISPATCH_NCHUNK(nc, rms, {
    if (rms) {
      if (has_res) NORM_FWD_LAUNCH(kNC, true, true);
      else NORM_FWD_LAUNCH(kNC, true, false);
    } else {
      if (has_res) NORM_FWD_LAUNCH(kNC, false, true);
      else NORM_FWD_LAUNCH(kNC, false, false);
    }
  });
#undef NORM_FWD_LAUNCH
  HIP_CHECK_LAST();
  return {y, has_res ? sum_out : x, mean.defined() ? mean : rstd, rstd};
}

std::vector<torch::Tensor> norm_bwd(torch::Tensor dy, torch::Tensor x,
                                    c10::optional<torch::Tensor> mean,
                                    torch::Tensor rstd, torch::Tensor gamma,
                                    bool rms,
                                    c10::optional<torch::Tensor> dadd) {
  check_norm_inputs(x, gamma);
  TORCH_CHECK(dy.is_contiguous() && dy.scalar_type() == torch::kBFloat16);
  const short* dadd_p = nullptr;
  if (dadd.has_value() && dadd->defined()) {
    TORCH_CHECK(dadd->is_contiguous() &&
                dadd->scalar_type() == torch::kBFloat16 &&
                dadd->numel() == x.numel());
    dadd_p = (const short*)dadd->data_ptr();
  }
  const int D = x.size(-1);
  const long n_rows = x.numel() / D;
  auto dx = torch::empty_like(x);
  int nc = pick_nchunk(D);
  int grid = (int)std::min<long>(n_rows, 512);
  if (D <= 1024)
    grid = (int)std::min<long>((n_rows + 3) / 4, 512);
  auto fopt = x.options().dtype(torch::kFloat32);
  // dgamma partials at column 0..D-1, dbeta partials (LN) at D..2D-1:
  // one buffer, one reduction pass for both.
  const long D2 = rms ? D : 2L * D;
  auto part = torch::empty({grid, D2}, fopt);
  float* part_dg_p = part.data_ptr<float>();
  float* part_db_p = rms ? nullptr : part_dg_p + D;
  auto stream = c10::cuda::getCurrentCUDAStream();
  if (D <= 1024) {
    if (rms) {
      hipLaunchKernelGGL((norm_bwd_wave_kernel<true>), dim3(grid),
                         dim3(NT), 2 * D * 4, stream.stream(),
                         (const short*)dy.data_ptr(),
                         (const short*)x.data_ptr(), nullptr,
                         rstd.data_ptr<float>(),
                         (const short*)gamma.data_ptr(), dadd_p,
                         (short*)dx.data_ptr(),
                         part_dg_p, nullptr, D2, n_rows, D);
    } else {
      hipLaunchKernelGGL((norm_bwd_wave_kernel<false>), dim3(grid),
                         dim3(NT), 2 * D * 4, stream.stream(),
                         (const short*)dy.data_ptr(),
                         (const short*)x.data_ptr(),
                         mean->data_ptr<float>(),
                         rstd.data_ptr<float>(),
                         (const short*)gamma.data_ptr(), dadd_p,
                         (short*)dx.data_ptr(),
                         part_dg_p, part_db_p, D2, n_rows, D);
    }
  } else
  DISPATCH_NCHUNK(nc, rms, {
    if (rms) {
      hipLaunchKernelGGL((norm_bwd_kernel<kNC, true>), dim3(grid), dim3(NT),
                         0, stream.stream(), (const short*)dy.data_ptr(),
                         (const short*)x.data_ptr(), nullptr,
                         rstd.data_ptr<float>(),
                         (const short*)gamma.data_ptr(), dadd_p,
                         (short*)dx.data_ptr(), part_dg_p,
                         nullptr, D2, (int)n_rows, D);
    } else {
      hipLaunchKernelGGL((norm_bwd_kernel<kNC, false>), dim3(grid), dim3(NT),
                         0, stream.stream(), (const short*)dy.data_ptr(),
                         (const short*)x.data_ptr(),
                         mean->data_ptr<float>(), rstd.data_ptr<float>(),
                         (const short*)gamma.data_ptr(), dadd_p,
                         (short*)dx.data_ptr(), part_dg_p,
                         part_db_p, D2, (int)n_rows, D);
    }
  });
  auto gb = torch::empty({D2}, x.options());  // [dgamma | dbeta] bf16
  hipLaunchKernelGGL(sum_cast_bf16_kernel,
                     dim3((int)((D2 + 63) / 64)), dim3(1024), 0,
                     stream.stream(), part.data_ptr<float>(),
                     (short*)gb.data_ptr(), grid, (int)D2);
  auto dgamma = gb.narrow(0, 0, D);
  HIP_CHECK_LAST();
  return {dx, dgamma, rms ? dgamma : gb.narrow(0, D, D)};
}
