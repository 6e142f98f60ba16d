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

// Multi-tensor AdamW: ONE launch updates every bf16 parameter of the
// optimizer (vs one launch per tensor).  A chunk table (built once —
// pointers are stable across steps and hipGraph replays) maps blocks
// to (tensor, offset) pieces; the kernel also ZEROES the consumed
// grad, so no zero_grad pass exists at all and grad tensors keep
// their identity for graph capture.
constexpr int MT_CHUNK = 8192;  // elements per table entry

// device-side step counter + bias corrections: host-passed values
// would be frozen by hipGraph capture; sb = [step, bc1, bc2] f32.
__global__ void adamw_mt_prestep_kernel(float* sb, float beta1,
                                        float beta2) {
  float st = sb[0] + 1.f;
  sb[0] = st;
  sb[1] = 1.f - powf(beta1, st);
  sb[2] = 1.f - powf(beta2, st);
}

__global__ __launch_bounds__(NT) void adamw_mt_kernel(
    const long* __restrict__ ptrs,   // [T][5]: p,p32,g,m,v
    const long* __restrict__ sizes,  // [T]
    const int* __restrict__ chunk_t, const long* __restrict__ chunk_off,
    int n_chunks, float lr, float beta1, float beta2, float eps,
    float wd, const float* __restrict__ sb) {
  const float bc1 = sb[1], bc2 = sb[2];
  for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    const int t = chunk_t[c];
    const long off = chunk_off[c];
    short* p = (short*)ptrs[t * 5 + 0] + off;
    float* p32 = (float*)ptrs[t * 5 + 1] + off;
    short* g = (short*)ptrs[t * 5 + 2] + off;
    float* m = (float*)ptrs[t * 5 + 3] + off;
    float* v = (float*)ptrs[t * 5 + 4] + off;
    const long n = min((long)MT_CHUNK, sizes[t] - off);
    // vectorized main path (16-B f32 / 8-B bf16 accesses; chunk bases
    // are MT_CHUNK-aligned so alignment holds except at tensor tails)
    for (long i = (long)threadIdx.x * 4; i < n; i += NT * 4) {
      if (i + 3 < n) {
        short4v gv = *reinterpret_cast<const short4v*>(g + i);
        float4v mp = *reinterpret_cast<const float4v*>(m + i);
        float4v vp = *reinterpret_cast<const float4v*>(v + i);
        float4v pp = *reinterpret_cast<const float4v*>(p32 + i);
        short4v pout, gz;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          float gg = bf16_to_f32(gv[j]);
          float mm = beta1 * mp[j] + (1.f - beta1) * gg;
          float vv = beta2 * vp[j] + (1.f - beta2) * gg * gg;
          float pw = pp[j];
          pw -= lr * ((mm / bc1) / (sqrtf(vv / bc2) + eps) + wd * pw);
          mp[j] = mm; vp[j] = vv; pp[j] = pw;
          pout[j] = f32_to_bf16(pw);
          gz[j] = 0;
        }
        *reinterpret_cast<float4v*>(m + i) = mp;
        *reinterpret_cast<float4v*>(v + i) = vp;
        *reinterpret_cast<float4v*>(p32 + i) = pp;
        *reinterpret_cast<short4v*>(p + i) = pout;
        *reinterpret_cast<short4v*>(g + i) = gz;  // fused zero_grad
      } else {
        for (long k = i; k < n; ++k) {
          float gg = bf16_to_f32(g[k]);
          float mm = beta1 * m[k] + (1.f - beta1) * gg;
          float vv = beta2 * v[k] + (1.f - beta2) * gg * gg;
          float pw = p32[k];
          pw -= lr * ((mm / bc1) / (sqrtf(vv / bc2) + eps) + wd * pw);
          m[k] = mm; v[k] = vv; p32[k] = pw;
          p[k] = f32_to_bf16(pw);
          g[k] = 0;
        }
      }
    }
  }
}

// Multi-tensor SGD-momentum: one launch per dtype group (bf16 params
// with fp32 masters; fp32 params e.g. BatchNorm affine).  Zeroes the
// consumed grads like the AdamW variant.
template <bool BF16>
__global__ __launch_bounds__(NT) void sgd_mt_kernel(
    const long* __restrict__ ptrs,   // [T][4]: p, p32(or 0), g, buf
    const long* __restrict__ sizes, const int* __restrict__ chunk_t,
    const long* __restrict__ chunk_off, int n_chunks, float lr,
    float momentum, float wd) {
  for (int c = blockIdx.x; c < n_chunks; c += gridDim.x) {
    const int t = chunk_t[c];
    const long off = chunk_off[c];
    const long n = min((long)MT_CHUNK, sizes[t] - off);
    float* buf = (float*)ptrs[t * 4 + 3] + off;
    if (BF16) {
      short* p = (short*)ptrs[t * 4 + 0] + off;
      float* p32 = (float*)ptrs[t * 4 + 1] + off;
      short* g = (short*)ptrs[t * 4 + 2] + off;
      for (long i = (long)threadIdx.x * 4; i < n; i += NT * 4) {
        long lim = min(i + 4, n);
        for (long k = i; k < lim; ++k) {
          float gg = bf16_to_f32(g[k]) + wd * p32[k];
          float b = momentum * buf[k] + gg;
          buf[k] = b;
          float pw = p32[k] - lr * b;
          p32[k] = pw;
          p[k] = f32_to_bf16(pw);
          g[k] = 0;
        }
      }
    } else {
      float* p = (float*)ptrs[t * 4 + 0] + off;
      float* g = (float*)ptrs[t * 4 + 2] + off;
      for (long i = (long)threadIdx.x * 4; i < n; i += NT * 4) {
        long lim = min(i + 4, n);
        for (long k = i; k < lim; ++k) {
          float gg = g[k] + wd * p[k];
          float b = momentum * buf[k] + gg;
          buf[k] = b;
          p[k] -= lr * b;
          g[k] = 0.f;
        }
      }
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

void adamw_multi(torch::Tensor ptrs, torch::Tensor sizes,
                 torch::Tensor chunk_t, torch::Tensor chunk_off,
                 torch::Tensor stepbuf, double lr, double beta1,
                 double beta2, double eps, double wd) {
  TORCH_CHECK(ptrs.is_cuda() && ptrs.scalar_type() == torch::kLong);
  int n_chunks = chunk_t.numel();
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(adamw_mt_prestep_kernel, dim3(1), dim3(1), 0,
                     stream.stream(), stepbuf.data_ptr<float>(),
                     (float)beta1, (float)beta2);
  int grid = std::min(n_chunks, 2048);
  hipLaunchKernelGGL(adamw_mt_kernel, dim3(grid), dim3(NT), 0,
                     stream.stream(), ptrs.data_ptr<long>(),
                     sizes.data_ptr<long>(), chunk_t.data_ptr<int>(),
                     chunk_off.data_ptr<long>(), n_chunks, (float)lr,
                     (float)beta1, (float)beta2, (float)eps, (float)wd,
                     stepbuf.data_ptr<float>());
  HIP_CHECK_LAST();
}

void sgd_multi(torch::Tensor ptrs, torch::Tensor sizes,
               torch::Tensor chunk_t, torch::Tensor chunk_off,
               double lr, double momentum, double wd, bool bf16) {
  TORCH_CHECK(ptrs.is_cuda() && ptrs.scalar_type() == torch::kLong);
  int n_chunks = chunk_t.numel();
  auto stream = c10::cuda::getCurrentCUDAStream();
  int grid = std::min(n_chunks, 2048);
  if (bf16) {
    hipLaunchKernelGGL((sgd_mt_kernel<true>), dim3(grid), dim3(NT), 0,
                       stream.stream(), ptrs.data_ptr<long>(),
                       sizes.data_ptr<long>(), chunk_t.data_ptr<int>(),
                       chunk_off.data_ptr<long>(), n_chunks, (float)lr,
                       (float)momentum, (float)wd);
  } else {
    hipLaunchKernelGGL((sgd_mt_kernel<false>), dim3(grid), dim3(NT), 0,
                       stream.stream(), ptrs.data_ptr<long>(),
                       sizes.data_ptr<long>(), chunk_t.data_ptr<int>(),
                       chunk_off.data_ptr<long>(), n_chunks, (float)lr,
                       (float)momentum, (float)wd);
  }
  HIP_CHECK_LAST();
}

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
