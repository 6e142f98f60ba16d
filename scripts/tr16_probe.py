"""Empirically determine ds_read_b64_tr_b16's lane gather semantics on
gfx950: fill LDS with identity (lds[i] = i), each lane reads 8 B at
base + lane*8, print which LDS indices land in each lane's 4 outputs.
"""
import os
import sys

import torch
from torch.utils.cpp_extension import load_inline

SRC = r"""
#include <torch/extension.h>
#include <hip/hip_runtime.h>
typedef __attribute__((ext_vector_type(4))) short short4v;

__global__ void tr_probe_kernel(const short* in, short* out, int mode) {
  __shared__ short lds[2048];
  int tid = threadIdx.x;
  for (int i = tid; i < 2048; i += blockDim.x) lds[i] = in[i];
  __syncthreads();
  unsigned base = (unsigned)(size_t)(&lds[0]);
  unsigned addr;
  if (mode == 0) addr = base + tid * 8;            // flat 8B/lane
  else if (mode == 1) addr = base + tid * 16;      // 16B stride
  else addr = base + ((tid & 15) + (tid >> 4) * 64) * 2;  // row guess
  short4v v;
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v) : "v"(addr));
#pragma unroll
  for (int j = 0; j < 4; ++j) out[tid * 4 + j] = v[j];
}

torch::Tensor tr_probe(torch::Tensor in, long mode) {
  auto out = torch::zeros({64 * 4}, in.options());
  hipLaunchKernelGGL(tr_probe_kernel, dim3(1), dim3(64), 0, 0,
                     (const short*)in.data_ptr(), (short*)out.data_ptr(),
                     (int)mode);
  return out;
}
PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) { m.def("tr_probe", &tr_probe); }
"""


def main():
    os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    mod = load_inline(name="tr_probe_ext", cpp_sources="",
                      cuda_sources=SRC, with_cuda=True, verbose=False)
    iden = torch.arange(2048, dtype=torch.int16, device="cuda")
    for mode in (0, 1, 2):
        out = mod.tr_probe(iden, mode).cpu().view(64, 4)
        print(f"--- mode {mode} (lane: 4 received LDS indices)")
        for lane in range(0, 64):
            print(f"  lane {lane:2d}: {out[lane].tolist()}")


if __name__ == "__main__":
    main()
