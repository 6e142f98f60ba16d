// Fused embedding lookup for gfx950 (bf16 weight/output).
//
// Why this exists: torch's nn.Embedding dense backward routes through
// rocprim radix-sort + partition/unique kernels.  Those are legal
// under hipGraph capture, but replaying them from many concurrently
// captured job graphs proved fragile on ROCm 7 (HSA aperture faults in
// the captured partition kernel after unrelated allocator traffic —
// see scripts/repro_prewarm*.py).  These kernels keep the whole
// embedding path in plain, capture-trivial code:
//
// * forward: one vectorized gather pass (short8 = 16 B per lane),
// * backward: fp32 workspace scatter-add (global atomicAdd — fp32
//   accumulation regardless of weight dtype; summation ORDER is
//   atomic-arrival, so the bf16 rounding of repeated tokens is not
//   bit-reproducible — same class as torch's atomic path), then one
//   vectorized fp32->bf16 cast pass into the dense grad.
#include <torch/extension.h>
#include <c10/cuda/CUDAStream.h>
#include "common.h"

namespace {

constexpr int NT = 256;

__global__ __launch_bounds__(NT) void embed_fwd_kernel(
    const short* __restrict__ W, const long* __restrict__ idx,
    short* __restrict__ out, long n, int H) {
  const long per_row = H / 8;
  const long total = n * per_row;
  long chunk = (long)blockIdx.x * NT + threadIdx.x;
  const long stride = (long)gridDim.x * NT;
  for (; chunk < total; chunk += stride) {
    const long row = chunk / per_row;
    const long h = (chunk % per_row) * 8;
    *reinterpret_cast<short8*>(out + row * H + h) =
        *reinterpret_cast<const short8*>(W + idx[row] * (long)H + h);
  }
}

__global__ __launch_bounds__(NT) void embed_bwd_scatter_kernel(
    const short* __restrict__ dy, const long* __restrict__ idx,
    float* __restrict__ ws, long n, int H) {
  // 2 elems per thread: atomicAdd is scalar fp32, but the dy load
  // stays 32-bit aligned
  const long per_row = H / 2;
  const long total = n * per_row;
  long e = (long)blockIdx.x * NT + threadIdx.x;
  const long stride = (long)gridDim.x * NT;
  for (; e < total; e += stride) {
    const long row = e / per_row;
    const long h = (e % per_row) * 2;
    const short* d = dy + row * H + h;
    float* w = ws + idx[row] * (long)H + h;
    atomicAdd(w, bf16_to_f32(d[0]));
    atomicAdd(w + 1, bf16_to_f32(d[1]));
  }
}

__global__ __launch_bounds__(NT) void f32_to_bf16_kernel(
    const float* __restrict__ src, short* __restrict__ dst, long total8) {
  long chunk = (long)blockIdx.x * NT + threadIdx.x;
  const long stride = (long)gridDim.x * NT;
  for (; chunk < total8; chunk += stride) {
    const float4v a = *reinterpret_cast<const float4v*>(src + chunk * 8);
    const float4v b =
        *reinterpret_cast<const float4v*>(src + chunk * 8 + 4);
    short8 o;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      o[j] = f32_to_bf16(a[j]);
      o[4 + j] = f32_to_bf16(b[j]);
    }
    *reinterpret_cast<short8*>(dst + chunk * 8) = o;
  }
}

}  // namespace

torch::Tensor embed_fwd(torch::Tensor weight, torch::Tensor idx) {
  TORCH_CHECK(weight.is_cuda() && weight.dim() == 2 &&
              weight.scalar_type() == torch::kBFloat16 &&
              weight.is_contiguous());
  TORCH_CHECK(idx.scalar_type() == torch::kLong);
  const int H = weight.size(1);
  TORCH_CHECK(H % 8 == 0, "hidden dim must be a multiple of 8");
  auto idxc = idx.contiguous();
  const long n = idxc.numel();
  auto sizes = idx.sizes().vec();
  sizes.push_back(H);
  auto out = torch::empty(sizes, weight.options());
  auto stream = c10::cuda::getCurrentCUDAStream();
  long total = n * (H / 8);
  int grid = (int)std::min<long>((total + NT - 1) / NT, 8192);
  hipLaunchKernelGGL(embed_fwd_kernel, dim3(grid), dim3(NT), 0,
                     stream.stream(), (const short*)weight.data_ptr(),
                     idxc.data_ptr<long>(), (short*)out.data_ptr(), n, H);
  HIP_CHECK_LAST();
  return out;
}

torch::Tensor embed_bwd(torch::Tensor dy, torch::Tensor idx, long V) {
  TORCH_CHECK(dy.is_cuda() && dy.scalar_type() == torch::kBFloat16);
  auto dyc = dy.contiguous();
  auto idxc = idx.contiguous();
  const int H = dy.size(-1);
  TORCH_CHECK(H % 8 == 0);
  const long n = idxc.numel();
  auto ws = torch::zeros({V, (long)H},
                         dy.options().dtype(torch::kFloat32));
  auto dw = torch::empty({V, (long)H}, dy.options());
  auto stream = c10::cuda::getCurrentCUDAStream();
  long total = n * (H / 2);
  int grid = (int)std::min<long>((total + NT - 1) / NT, 8192);
  hipLaunchKernelGGL(embed_bwd_scatter_kernel, dim3(grid), dim3(NT), 0,
                     stream.stream(), (const short*)dyc.data_ptr(),
                     idxc.data_ptr<long>(), ws.data_ptr<float>(), n, H);
  long total8 = V * (long)H / 8;
  int grid2 = (int)std::min<long>((total8 + NT - 1) / NT, 8192);
  hipLaunchKernelGGL(f32_to_bf16_kernel, dim3(grid2), dim3(NT), 0,
                     stream.stream(), ws.data_ptr<float>(),
                     (short*)dw.data_ptr(), total8);
  HIP_CHECK_LAST();
  return dw;
}
