// Fused cross-entropy over bf16 logits for gfx950.
//
// torch's path materializes fp32 logits (logits.float(): +6 bytes/elem
// traffic), then runs separate softmax + nll kernels.  Here:
// forward = ONE online-softmax pass over the bf16 logits per row
// (running max/sum in registers, wave+block combine), saving only the
// per-row logsumexp; backward = ONE elementwise pass producing bf16
// dlogits = (softmax - onehot) * gscale.  Vocab-sized fp32 tensors
// never exist.
#include <torch/extension.h>
#include <c10/cuda/CUDAStream.h>
#include "common.h"

namespace {

constexpr int NT = 256;

// combine two (m, s) softmax partials
__device__ __forceinline__ void sm_combine(float& m, float& s, float m2,
                                           float s2) {
  float mn = fmaxf(m, m2);
  s = s * __expf(m - mn) + s2 * __expf(m2 - mn);
  m = mn;
}

__global__ __launch_bounds__(NT) void ce_fwd_kernel(
    const short* __restrict__ logits, const long* __restrict__ labels,
    float* __restrict__ lse_out, float* __restrict__ loss_out,
    int n_rows, int V, float inv_n) {
  __shared__ float sm[2][NT / WAVE];
  for (int row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const short* xr = logits + (long)row * V;
    float m = -1e30f, s = 0.f;
    int i = threadIdx.x * 8;
    for (; i + 7 < V; i += NT * 8) {
      short8 v = *reinterpret_cast<const short8*>(xr + i);
      float loc_m = -1e30f;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        loc_m = fmaxf(loc_m, bf16_to_f32(v[j]));
      float loc_s = 0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        loc_s += __expf(bf16_to_f32(v[j]) - loc_m);
      sm_combine(m, s, loc_m, loc_s);
    }
    for (i = (V / 8) * 8 + threadIdx.x; i < V; i += NT)
      sm_combine(m, s, bf16_to_f32(xr[i]), 1.0f);
    // wave reduce
#pragma unroll
    for (int off = 32; off > 0; off >>= 1)
      sm_combine(m, s, __shfl_xor(m, off, WAVE),
                 __shfl_xor(s, off, WAVE));
    const int wid = threadIdx.x / WAVE;
    if ((threadIdx.x & 63) == 0) {
      sm[0][wid] = m;
      sm[1][wid] = s;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      float bm = sm[0][0], bs = sm[1][0];
      for (int w = 1; w < NT / WAVE; ++w)
        sm_combine(bm, bs, sm[0][w], sm[1][w]);
      float lse = bm + __logf(bs);
      lse_out[row] = lse;
      float xl = bf16_to_f32(xr[labels[row]]);
      atomicAdd(loss_out, (lse - xl) * inv_n);
    }
    __syncthreads();
  }
}

__global__ __launch_bounds__(NT) void ce_bwd_kernel(
    const short* __restrict__ logits, const long* __restrict__ labels,
    const float* __restrict__ lse, const float* __restrict__ gscale,
    short* __restrict__ dx, long n_rows, int V) {
  long chunk = (long)blockIdx.x * NT + threadIdx.x;
  long total = n_rows * (long)((V + 7) / 8);
  long stride = (long)gridDim.x * NT;
  const float g = gscale[0];
  for (; chunk < total; chunk += stride) {
    long row = chunk / ((V + 7) / 8);
    int i = (int)(chunk % ((V + 7) / 8)) * 8;
    const short* xr = logits + row * V;
    short* dr = dx + row * V;
    float l = lse[row];
    long lab = labels[row];
    if (i + 7 < V) {
      short8 v = *reinterpret_cast<const short8*>(xr + i);
      short8 out;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float p = __expf(bf16_to_f32(v[j]) - l);
        out[j] = f32_to_bf16((p - (i + j == lab ? 1.f : 0.f)) * g);
      }
      *reinterpret_cast<short8*>(dr + i) = out;
    } else {
      for (int j = 0; i + j < V; ++j) {
        float p = __expf(bf16_to_f32(xr[i + j]) - l);
        dr[i + j] = f32_to_bf16((p - (i + j == lab ? 1.f : 0.f)) * g);
      }
    }
  }
}

}  // namespace

std::vector<torch::Tensor> ce_fwd(torch::Tensor logits,
                                  torch::Tensor labels) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2 &&
              logits.scalar_type() == torch::kBFloat16 &&
              logits.is_contiguous());
  TORCH_CHECK(labels.scalar_type() == torch::kLong);
  const long n = logits.size(0);
  const int V = logits.size(1);
  auto fopt = logits.options().dtype(torch::kFloat32);
  auto lse = torch::empty({n}, fopt);
  auto loss = torch::zeros({}, fopt);
  auto stream = c10::cuda::getCurrentCUDAStream();
  int grid = (int)std::min<long>(n, 4096);
  hipLaunchKernelGGL(ce_fwd_kernel, dim3(grid), dim3(NT), 0,
                     stream.stream(), (const short*)logits.data_ptr(),
                     labels.data_ptr<long>(), lse.data_ptr<float>(),
                     loss.data_ptr<float>(), (int)n, V, 1.0f / (float)n);
  HIP_CHECK_LAST();
  return {loss, lse};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor labels,
                     torch::Tensor lse, torch::Tensor gscale) {
  const long n = logits.size(0);
  const int V = logits.size(1);
  auto dx = torch::empty_like(logits);
  auto stream = c10::cuda::getCurrentCUDAStream();
  long total = n * ((V + 7) / 8);
  int grid = (int)std::min<long>((total + NT - 1) / NT, 4096);
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(grid), dim3(NT), 0,
                     stream.stream(), (const short*)logits.data_ptr(),
                     labels.data_ptr<long>(), lse.data_ptr<float>(),
                     gscale.data_ptr<float>(), (short*)dx.data_ptr(),
                     n, V);
  HIP_CHECK_LAST();
  return dx;
}
