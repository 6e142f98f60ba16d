// Fused elementwise kernels for the Llama path on gfx950:
//  * RoPE (rotate-half) forward/backward from a host-precomputed
//    fp32 cos/sin table (guide: no on-device trig) — replaces ~8
//    torch ops (chunk/mul/cat) per projection per direction.
//  * SwiGLU gate: y = silu(g) * u, with fused backward — replaces the
//    silu/mul kernel pair and their backward chain.
// All memory-bound: bf16x8 vectorized, grid-stride.
#include <torch/extension.h>
#include <c10/cuda/CUDAStream.h>
#include "common.h"

namespace {

constexpr int NT = 256;

// x [rows, D] where row r maps to sequence position pos = r % S
// (rows are [B,H,S] flattened).  cos/sin: [S, D/2] fp32.
// STRIDED: the input may be a [B,H,S,D] VIEW (e.g. a head slice of
// the packed qkv projection transposed to BHSD) read through a
// (b,h,s) stride triple — no .contiguous() materialization; the
// OUTPUT is always written contiguous [B,H,S,D] (what the attention
// kernels consume).
template <bool BWD, bool STRIDED>
__global__ __launch_bounds__(NT) void rope_kernel(
    const short* __restrict__ x, const float* __restrict__ cosb,
    const float* __restrict__ sinb, short* __restrict__ out,
    long rows, int S, int D, int Hh, long sB, long sH, long sS) {
  const int HD = D / 2;
  const int chunks = HD / 8;  // 8-wide pieces per half-row
  long total = rows * chunks;
  long i = (long)blockIdx.x * NT + threadIdx.x;
  long stride = (long)gridDim.x * NT;
  for (; i < total; i += stride) {
    long row = i / chunks;
    int c8 = (int)(i % chunks) * 8;
    int pos = (int)(row % S);
    long xoff;
    if (STRIDED) {
      long bh = row / S;
      xoff = (bh / Hh) * sB + (bh % Hh) * sH + (long)pos * sS;
    } else {
      xoff = row * D;
    }
    const short* xr = x + xoff;
    short* orow = out + row * D;  // output always contiguous
    short8 x1 = *reinterpret_cast<const short8*>(xr + c8);
    short8 x2 = *reinterpret_cast<const short8*>(xr + HD + c8);
    const float* cr = cosb + (long)pos * HD + c8;
    const float* sr = sinb + (long)pos * HD + c8;
    short8 o1, o2;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float a = bf16_to_f32(x1[j]), b = bf16_to_f32(x2[j]);
      float cc = cr[j], ss = sr[j];
      if (!BWD) {
        o1[j] = f32_to_bf16(a * cc - b * ss);
        o2[j] = f32_to_bf16(b * cc + a * ss);
      } else {  // transpose of the rotation
        o1[j] = f32_to_bf16(a * cc + b * ss);
        o2[j] = f32_to_bf16(b * cc - a * ss);
      }
    }
    *reinterpret_cast<short8*>(orow + c8) = o1;
    *reinterpret_cast<short8*>(orow + HD + c8) = o2;
  }
}

__global__ __launch_bounds__(NT) void silu_mul_fwd_kernel(
    const short* __restrict__ g, const short* __restrict__ u,
    short* __restrict__ y, long n) {
  long i = ((long)blockIdx.x * NT + threadIdx.x) * 8;
  long stride = (long)gridDim.x * NT * 8;
  for (; i + 7 < n; i += stride) {
    short8 gv = *reinterpret_cast<const short8*>(g + i);
    short8 uv = *reinterpret_cast<const short8*>(u + i);
    short8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gg = bf16_to_f32(gv[j]);
      float sig = 1.f / (1.f + __expf(-gg));
      out[j] = f32_to_bf16(gg * sig * bf16_to_f32(uv[j]));
    }
    *reinterpret_cast<short8*>(y + i) = out;
  }
}

__global__ __launch_bounds__(NT) void silu_mul_bwd_kernel(
    const short* __restrict__ dy, const short* __restrict__ g,
    const short* __restrict__ u, short* __restrict__ dg,
    short* __restrict__ du, long n) {
  long i = ((long)blockIdx.x * NT + threadIdx.x) * 8;
  long stride = (long)gridDim.x * NT * 8;
  for (; i + 7 < n; i += stride) {
    short8 dv = *reinterpret_cast<const short8*>(dy + i);
    short8 gv = *reinterpret_cast<const short8*>(g + i);
    short8 uv = *reinterpret_cast<const short8*>(u + i);
    short8 og, ou;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float d = bf16_to_f32(dv[j]);
      float gg = bf16_to_f32(gv[j]);
      float sig = 1.f / (1.f + __expf(-gg));
      float si = gg * sig;
      ou[j] = f32_to_bf16(d * si);
      og[j] = f32_to_bf16(d * bf16_to_f32(uv[j]) *
                          (sig * (1.f + gg * (1.f - sig))));
    }
    *reinterpret_cast<short8*>(dg + i) = og;
    *reinterpret_cast<short8*>(du + i) = ou;
  }
}


// tanh-approximate GELU (matches torch F.gelu(approximate="tanh")):
// y = 0.5 x (1 + tanh(c(x + a x^3))), c = sqrt(2/pi), a = 0.044715.
// One fused pass each way — replaces torch's separate Gelu{,Backward}
// elementwise kernels in the BERT FFN.
__global__ __launch_bounds__(NT) void gelu_fwd_kernel(
    const short* __restrict__ x, short* __restrict__ y, long n) {
  constexpr float C = 0.7978845608028654f, A = 0.044715f;
  long i = ((long)blockIdx.x * NT + threadIdx.x) * 8;
  long stride = (long)gridDim.x * NT * 8;
  for (; i + 7 < n; i += stride) {
    short8 xv = *reinterpret_cast<const short8*>(x + i);
    short8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = bf16_to_f32(xv[j]);
      float t = tanhf(C * (v + A * v * v * v));
      out[j] = f32_to_bf16(0.5f * v * (1.f + t));
    }
    *reinterpret_cast<short8*>(y + i) = out;
  }
}

__global__ __launch_bounds__(NT) void gelu_bwd_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    short* __restrict__ dx, long n) {
  constexpr float C = 0.7978845608028654f, A = 0.044715f;
  long i = ((long)blockIdx.x * NT + threadIdx.x) * 8;
  long stride = (long)gridDim.x * NT * 8;
  for (; i + 7 < n; i += stride) {
    short8 dv = *reinterpret_cast<const short8*>(dy + i);
    short8 xv = *reinterpret_cast<const short8*>(x + i);
    short8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float v = bf16_to_f32(xv[j]);
      float t = tanhf(C * (v + A * v * v * v));
      float du = C * (1.f + 3.f * A * v * v);
      float g = 0.5f * (1.f + t) + 0.5f * v * (1.f - t * t) * du;
      out[j] = f32_to_bf16(bf16_to_f32(dv[j]) * g);
    }
    *reinterpret_cast<short8*>(dx + i) = out;
  }
}


// Packed SwiGLU: gu = [rows][2F] (one fused gate_up projection);
// y = silu(gu[:, :F]) * gu[:, F:].  The packed forms read the
// projection's halves through the row stride and the backward writes
// BOTH grads into one [rows][2F] buffer — no contiguous() copies in,
// no chunk-backward cat out.
__global__ __launch_bounds__(NT) void swiglu_fwd_kernel(
    const short* __restrict__ gu, short* __restrict__ y,
    long rows, int F) {
  const long per_row = F / 8;
  const long total = rows * per_row;
  long idx = (long)blockIdx.x * NT + threadIdx.x;
  const long stride = (long)gridDim.x * NT;
  for (; idx < total; idx += stride) {
    const long r = idx / per_row;
    const int c = (int)(idx % per_row) * 8;
    const short* base = gu + r * 2 * F;
    short8 gv = *reinterpret_cast<const short8*>(base + c);
    short8 uv = *reinterpret_cast<const short8*>(base + F + c);
    short8 out;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float gg = bf16_to_f32(gv[j]);
      float sig = 1.f / (1.f + __expf(-gg));
      out[j] = f32_to_bf16(gg * sig * bf16_to_f32(uv[j]));
    }
    *reinterpret_cast<short8*>(y + r * F + c) = out;
  }
}

__global__ __launch_bounds__(NT) void swiglu_bwd_kernel(
    const short* __restrict__ dy, const short* __restrict__ gu,
    short* __restrict__ dgu, long rows, int F) {
  const long per_row = F / 8;
  const long total = rows * per_row;
  long idx = (long)blockIdx.x * NT + threadIdx.x;
  const long stride = (long)gridDim.x * NT;
  for (; idx < total; idx += stride) {
    const long r = idx / per_row;
    const int c = (int)(idx % per_row) * 8;
    const short* base = gu + r * 2 * F;
    short8 dv = *reinterpret_cast<const short8*>(dy + r * F + c);
    short8 gv = *reinterpret_cast<const short8*>(base + c);
    short8 uv = *reinterpret_cast<const short8*>(base + F + c);
    short8 og, ou;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float d = bf16_to_f32(dv[j]);
      float gg = bf16_to_f32(gv[j]);
      float sig = 1.f / (1.f + __expf(-gg));
      ou[j] = f32_to_bf16(d * gg * sig);
      og[j] = f32_to_bf16(d * bf16_to_f32(uv[j]) *
                          (sig * (1.f + gg * (1.f - sig))));
    }
    short* dbase = dgu + r * 2 * F;
    *reinterpret_cast<short8*>(dbase + c) = og;
    *reinterpret_cast<short8*>(dbase + F + c) = ou;
  }
}

int grid_for_n(long items) {
  return (int)std::min<long>((items + NT - 1) / NT, 2048);
}

}  // namespace

torch::Tensor rope_apply(torch::Tensor x, torch::Tensor cosb,
                         torch::Tensor sinb, bool backward) {
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(cosb.scalar_type() == torch::kFloat32);
  const int D = x.size(-1);
  const int S = x.size(-2);
  TORCH_CHECK(D % 16 == 0, "rope: D must be a multiple of 16");
  TORCH_CHECK(cosb.size(-1) == D / 2 && cosb.size(0) >= S);
  long rows = x.numel() / D;
  const bool cont = x.is_contiguous();
  int Hh = 1;
  long sB = 0, sH = 0, sS = 0;
  if (!cont) {
    TORCH_CHECK(x.dim() == 4 && x.stride(3) == 1 &&
                x.stride(2) % 8 == 0,
                "rope: strided input must be a [B,H,S,D] view with "
                "contiguous 16B-aligned D rows");
    Hh = x.size(1);
    sB = x.stride(0);
    sH = x.stride(1);
    sS = x.stride(2);
  }
  auto out = cont ? torch::empty_like(x)
                  : torch::empty(x.sizes(), x.options());
  long total = rows * (D / 16);
  auto stream = c10::cuda::getCurrentCUDAStream();
#define ROPE_LAUNCH(BWD, SR)                                            \
  hipLaunchKernelGGL((rope_kernel<BWD, SR>), dim3(grid_for_n(total)),   \
                     dim3(NT), 0, stream.stream(),                      \
                     (const short*)x.data_ptr(),                        \
                     cosb.data_ptr<float>(), sinb.data_ptr<float>(),    \
                     (short*)out.data_ptr(), rows, S, D, Hh, sB, sH, sS)
  if (backward) {
    if (cont) ROPE_LAUNCH(true, false);
    else ROPE_LAUNCH(true, true);
  } else {
    if (cont) ROPE_LAUNCH(false, false);
    else ROPE_LAUNCH(false, true);
  }
#undef ROPE_LAUNCH
  HIP_CHECK_LAST();
  return out;
}

torch::Tensor silu_mul_fwd(torch::Tensor g, torch::Tensor u) {
  TORCH_CHECK(g.is_cuda() && g.is_contiguous() && u.is_contiguous());
  TORCH_CHECK(g.scalar_type() == torch::kBFloat16);
  long n = g.numel();
  TORCH_CHECK(n % 8 == 0);
  auto y = torch::empty_like(g);
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(silu_mul_fwd_kernel, dim3(grid_for_n(n / 8)),
                     dim3(NT), 0, stream.stream(),
                     (const short*)g.data_ptr(),
                     (const short*)u.data_ptr(), (short*)y.data_ptr(),
                     n);
  HIP_CHECK_LAST();
  return y;
}

std::vector<torch::Tensor> silu_mul_bwd(torch::Tensor dy, torch::Tensor g,
                                        torch::Tensor u) {
  long n = g.numel();
  auto dg = torch::empty_like(g);
  auto du = torch::empty_like(u);
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(silu_mul_bwd_kernel, dim3(grid_for_n(n / 8)),
                     dim3(NT), 0, stream.stream(),
                     (const short*)dy.data_ptr(),
                     (const short*)g.data_ptr(),
                     (const short*)u.data_ptr(), (short*)dg.data_ptr(),
                     (short*)du.data_ptr(), n);
  HIP_CHECK_LAST();
  return {dg, du};
}

torch::Tensor gelu_fwd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == torch::kBFloat16);
  long n = x.numel();
  TORCH_CHECK(n % 8 == 0);
  auto y = torch::empty_like(x);
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(gelu_fwd_kernel, dim3(grid_for_n(n / 8)), dim3(NT),
                     0, stream.stream(), (const short*)x.data_ptr(),
                     (short*)y.data_ptr(), n);
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor gelu_bwd(torch::Tensor dy, torch::Tensor x) {
  long n = x.numel();
  auto dx = torch::empty_like(x);
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(gelu_bwd_kernel, dim3(grid_for_n(n / 8)), dim3(NT),
                     0, stream.stream(), (const short*)dy.data_ptr(),
                     (const short*)x.data_ptr(), (short*)dx.data_ptr(), n);
  HIP_CHECK_LAST();
  return dx;
}

torch::Tensor swiglu_fwd(torch::Tensor gu) {
  TORCH_CHECK(gu.is_cuda() && gu.is_contiguous() && gu.dim() >= 2 &&
              gu.scalar_type() == torch::kBFloat16);
  const int F2 = gu.size(-1);
  TORCH_CHECK(F2 % 16 == 0, "swiglu: last dim must be 2F, F % 8 == 0");
  const int F = F2 / 2;
  long rows = gu.numel() / F2;
  auto sizes = gu.sizes().vec();
  sizes.back() = F;
  auto y = torch::empty(sizes, gu.options());
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(grid_for_n(rows * (F / 8))),
                     dim3(NT), 0, stream.stream(),
                     (const short*)gu.data_ptr(), (short*)y.data_ptr(),
                     rows, F);
  HIP_CHECK_LAST();
  return y;
}

torch::Tensor swiglu_bwd(torch::Tensor dy, torch::Tensor gu) {
  const int F = gu.size(-1) / 2;
  long rows = gu.numel() / (2 * F);
  auto dgu = torch::empty_like(gu);
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(grid_for_n(rows * (F / 8))),
                     dim3(NT), 0, stream.stream(),
                     (const short*)dy.data_ptr(),
                     (const short*)gu.data_ptr(),
                     (short*)dgu.data_ptr(), rows, F);
  HIP_CHECK_LAST();
  return dgu;
}
