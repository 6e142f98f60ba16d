// Fused optimizer steps for gfx950: AdamW (bf16 param + fp32 master +
// fp32 m/v) and SGD-momentum (fp32 master + bf16 working copy).
//
// Memory-bound elementwise kernels: grid-stride, float4 / short8
// vectorized, one pass over all state (instead of torch's unfused
// 6-10 kernel launches + intermediate traffic per tensor).
#include <torch/extension.h>
#include <c10/cuda/CUDAStream.h>
#include "common.h"

namespace {

constexpr int NT = 256;

__global__ __launch_bounds__(NT) void adamw_kernel(
    short* __restrict__ p_bf16, float* __restrict__ p32,
    const short* __restrict__ grad, float* __restrict__ m,
    float* __restrict__ v, long n, float lr, float beta1, float beta2,
    float eps, float wd, float bc1, float bc2) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (long i = i0; i < n; i += stride) {
    if (i + 3 < n) {
      short4v gv = *reinterpret_cast<const short4v*>(grad + i);
      float4v mp = *reinterpret_cast<float4v*>(m + i);
      float4v vp = *reinterpret_cast<float4v*>(v + i);
      float4v pp = *reinterpret_cast<float4v*>(p32 + i);
      short4v pout;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float g = bf16_to_f32(gv[j]);
        float mm = beta1 * mp[j] + (1.f - beta1) * g;
        float vv = beta2 * vp[j] + (1.f - beta2) * g * g;
        float mhat = mm / bc1;
        float vhat = vv / bc2;
        float pw = pp[j];
        pw -= lr * (mhat / (sqrtf(vhat) + eps) + wd * pw);
        mp[j] = mm; vp[j] = vv; pp[j] = pw;
        pout[j] = f32_to_bf16(pw);
      }
      *reinterpret_cast<float4v*>(m + i) = mp;
      *reinterpret_cast<float4v*>(v + i) = vp;
      *reinterpret_cast<float4v*>(p32 + i) = pp;
      *reinterpret_cast<short4v*>(p_bf16 + i) = pout;
    } else {
      for (long k = i; k < n; ++k) {
        float g = bf16_to_f32(grad[k]);
        float mm = beta1 * m[k] + (1.f - beta1) * g;
        float vv = beta2 * v[k] + (1.f - beta2) * g * g;
        float pw = p32[k];
        pw -= lr * ((mm / bc1) / (sqrtf(vv / bc2) + eps) + wd * pw);
        m[k] = mm; v[k] = vv; p32[k] = pw;
        p_bf16[k] = f32_to_bf16(pw);
      }
    }
  }
}

__global__ __launch_bounds__(NT) void sgd_kernel(
    float* __restrict__ p32, short* __restrict__ p_bf16,
    const short* __restrict__ grad, float* __restrict__ buf, long n,
    float lr, float momentum, float wd) {
  long i0 = (long)(blockIdx.x * blockDim.x + threadIdx.x) * 4;
  long stride = (long)gridDim.x * blockDim.x * 4;
  for (long i = i0; i < n; i += stride) {
    long lim = (i + 4 < n) ? i + 4 : n;
    for (long k = i; k < lim; ++k) {
      float g = bf16_to_f32(grad[k]) + wd * p32[k];
      float b = momentum * buf[k] + g;
      buf[k] = b;
      float pw = p32[k] - lr * b;
      p32[k] = pw;
      p_bf16[k] = f32_to_bf16(pw);
    }
  }
}

// fp32-native SGD (BatchNorm affine params train in fp32)
__global__ __launch_bounds__(NT) void sgd_fp32_kernel(
    float* __restrict__ p, const float* __restrict__ grad,
    float* __restrict__ buf, long n, float lr, float momentum,
    float wd) {
  long i = blockIdx.x * blockDim.x + threadIdx.x;
  long stride = (long)gridDim.x * blockDim.x;
  for (; i < n; i += stride) {
    float g = grad[i] + wd * p[i];
    float b = momentum * buf[i] + g;
    buf[i] = b;
    p[i] -= lr * b;
  }
}

int grid_for(long n) {
  long blocks = (n / 4 + NT - 1) / NT;
  return (int)std::min<long>(blocks > 0 ? blocks : 1, 2048);
}

}  // namespace

void sgd_step_fp32(torch::Tensor p, torch::Tensor grad, torch::Tensor buf,
                   double lr, double momentum, double wd) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == torch::kFloat32);
  long n = p.numel();
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(sgd_fp32_kernel,
                     dim3((int)std::min<long>((n + NT - 1) / NT, 2048)),
                     dim3(NT), 0, stream.stream(), p.data_ptr<float>(),
                     grad.data_ptr<float>(), buf.data_ptr<float>(), n,
                     (float)lr, (float)momentum, (float)wd);
  HIP_CHECK_LAST();
}

void adamw_step(torch::Tensor p_bf16, torch::Tensor p32, torch::Tensor grad,
                torch::Tensor m, torch::Tensor v, double lr, double beta1,
                double beta2, double eps, double wd, long step) {
  TORCH_CHECK(p_bf16.is_cuda() && p_bf16.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(grad.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(p32.scalar_type() == torch::kFloat32);
  long n = p_bf16.numel();
  TORCH_CHECK(grad.numel() == n && m.numel() == n && v.numel() == n &&
              p32.numel() == n);
  float bc1 = 1.f - powf((float)beta1, (float)step);
  float bc2 = 1.f - powf((float)beta2, (float)step);
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(adamw_kernel, dim3(grid_for(n)), dim3(NT), 0,
                     stream.stream(), (short*)p_bf16.data_ptr(),
                     p32.data_ptr<float>(), (const short*)grad.data_ptr(),
                     m.data_ptr<float>(), v.data_ptr<float>(), n, (float)lr,
                     (float)beta1, (float)beta2, (float)eps, (float)wd,
                     bc1, bc2);
  HIP_CHECK_LAST();
}

void sgd_step(torch::Tensor p32, torch::Tensor p_bf16, torch::Tensor grad,
              torch::Tensor buf, double lr, double momentum, double wd) {
  TORCH_CHECK(p32.is_cuda() && p32.scalar_type() == torch::kFloat32);
  long n = p32.numel();
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(sgd_kernel, dim3(grid_for(n)), dim3(NT), 0,
                     stream.stream(), p32.data_ptr<float>(),
                     (short*)p_bf16.data_ptr(),
                     (const short*)grad.data_ptr(), buf.data_ptr<float>(),
                     n, (float)lr, (float)momentum, (float)wd);
  HIP_CHECK_LAST();
}
