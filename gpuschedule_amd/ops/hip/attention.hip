// Flash-attention forward for gfx950 (MI355X): online-softmax MFMA.
//
// Layout strategy (all fragments reuse the 16x16x32 bf16 MFMA mappings
// verified by the GEMM kernel):
//  * Q stays in registers: each wave owns 16 query rows; A-fragments
//    read straight from global (row = lane&15, 8 bf16 along D).
//  * K tile [KVB=64][D] staged in LDS with the GEMM XOR swizzle; QK^T
//    B-fragments are the same addressing as the GEMM B operand.
//  * V is PRE-TRANSPOSED once per call (fast LDS-tiled transpose2d
//    kernel) to [B,H,D,S], so the PV B-operand also reads 8 bf16
//    contiguous along the reduction (kv) dim — no ds transpose-read
//    gymnastics in v1.
//  * P (the probability tile) round-trips through a wave-private
//    XOR-swizzled LDS buffer to convert the MFMA C-layout into the
//    A-fragment layout.
//  * Online softmax: per-lane running m/l for 4 rows; row reductions
//    are 4 in-register values + a 16-lane xor-shuffle tree.
//
// One workgroup = 4 independent waves = 64 query rows of one (b, h).
// Causal masking: whole KV tiles after the diagonal are skipped;
// diagonal tiles mask per element.  Sequence tails are masked to -inf.
#include <torch/extension.h>
#include <c10/cuda/CUDAStream.h>
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

constexpr int KVB = 64;       // kv tile
constexpr int NTHREADS = 256; // 4 waves x 16 q rows

__device__ __forceinline__ int swz(int row, int byte_in_row, int rowb) {
  return row * rowb + (byte_in_row ^ ((row & 7) << 4));
}

// 16-lane-group reduction (rows live in groups of 16 lanes).
__device__ __forceinline__ float group_max(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}
__device__ __forceinline__ float group_sum(float v) {
#pragma unroll
  for (int off = 8; off > 0; off >>= 1)
    v += __shfl_xor(v, off, WAVE);
  return v;
}

template <int D, bool CAUSAL>
__global__ __launch_bounds__(NTHREADS, 2) void attn_fwd_kernel(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ Vt, short* __restrict__ O,
    int B, int H, int S, float scale) {
  constexpr int DF = D / 16;       // PV col fragments
  constexpr int KC = D / 32;       // QK^T k-steps over D
  constexpr int ROWB = D * 2;      // K tile LDS row bytes
  // LDS: K tile [KVB][D] + Vt tile [D][KVB] + 4 wave-private P tiles
  __shared__ __attribute__((aligned(16))) char lds[
      KVB * D * 2 + D * KVB * 2 + 4 * 16 * KVB * 2];
  char* k_lds = lds;
  char* v_lds = lds + KVB * D * 2;
  char* p_lds = lds + 2 * KVB * D * 2;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int tid = threadIdx.x;
  char* my_p = p_lds + wid * 16 * KVB * 2;

  const int bh = blockIdx.x / ((S + 63) / 64);
  const int qt = blockIdx.x % ((S + 63) / 64);
  const int b = bh / H, h = bh % H;
  const int q0 = qt * 64 + wid * 16;      // this wave's first q row

  const long qoff = ((long)(b * H + h) * S) * D;
  const long voff = ((long)(b * H + h) * D) * S;  // Vt is [D][S]

  // ---- Q into registers (A-fragment layout), pre-scaled ----------
  bf16x8 qreg[KC];
  {
    int row = q0 + (lane & 15);
    bool ok = row < S;
#pragma unroll
    for (int c = 0; c < KC; ++c) {
      int d0 = c * 32 + (lane >> 4) * 8;
      if (ok) {
        bf16x8 v = *reinterpret_cast<const bf16x8*>(
            Q + qoff + (long)row * D + d0);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          v[j] = (__bf16)((float)v[j] * scale);
        qreg[c] = v;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) qreg[c][j] = (__bf16)0.f;
      }
    }
  }

  float m[4], l[4];
  float4v o_acc[DF];
#pragma unroll
  for (int r = 0; r < 4; ++r) { m[r] = -1e30f; l[r] = 0.f; }
#pragma unroll
  for (int f = 0; f < DF; ++f) o_acc[f] = {0.f, 0.f, 0.f, 0.f};

  const int q_tile_end = qt * 64 + 63;
  const int n_kv = CAUSAL
      ? min((q_tile_end / KVB) + 1, (S + KVB - 1) / KVB)
      : (S + KVB - 1) / KVB;

  for (int t = 0; t < n_kv; ++t) {
    const int kv0 = t * KVB;
    // ---- stage K tile [KVB][D] and Vt tile [D][KVB] -------------
    {
      // K: KVB*D*2 bytes, 16B pieces: piece p -> row r, chunk c
      constexpr int PIECES_K = KVB * D * 2 / (NTHREADS * 16);
      constexpr int CPR = D * 2 / 16;   // 16B chunks per K row
#pragma unroll
      for (int p = 0; p < PIECES_K; ++p) {
        int idx = p * NTHREADS + tid;
        int r = idx / CPR, cb = (idx % CPR) * 16;
        int kvr = kv0 + r;
        short8 val;
        if (kvr < S) {
          val = *reinterpret_cast<const short8*>(
              K + qoff + (long)kvr * D + cb / 2);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) val[j] = 0;
        }
        *reinterpret_cast<short8*>(k_lds + swz(r, cb, ROWB)) = val;
      }
      // Vt: D rows x KVB cols
      constexpr int PIECES_V = D * KVB * 2 / (NTHREADS * 16);
      constexpr int CPRV = KVB * 2 / 16;
#pragma unroll
      for (int p = 0; p < PIECES_V; ++p) {
        int idx = p * NTHREADS + tid;
        int r = idx / CPRV, cb = (idx % CPRV) * 16;
        short8 val;
        int kvc = kv0 + cb / 2;
        if (kvc + 7 < S) {
          val = *reinterpret_cast<const short8*>(
              Vt + voff + (long)r * S + kvc);
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            val[j] = (kvc + j < S)
                ? (Vt + voff + (long)r * S)[kvc + j] : (short)0;
        }
        *reinterpret_cast<short8*>(
            v_lds + swz(r, cb, KVB * 2)) = val;
      }
    }
    __syncthreads();

    // ---- S = QK^T: 4 col-frags x KC k-steps ---------------------
    float4v s_acc[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) s_acc[f] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int kk = 0; kk < KC; ++kk) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 kf = *reinterpret_cast<const bf16x8*>(
            k_lds + swz(f * 16 + (lane & 15),
                        kk * 64 + (lane >> 4) * 16, ROWB));
        s_acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qreg[kk], kf, s_acc[f], 0, 0, 0);
      }
    }

    // ---- mask + online softmax ----------------------------------
    // lane holds S[row=(lane>>4)*4+r][col=f*16+(lane&15)]
    const int lrow0 = q0 + (lane >> 4) * 4;
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int col = kv0 + f * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        bool dead = col >= S;
        if (CAUSAL) dead |= col > lrow0 + r;
        if (dead) s_acc[f][r] = -1e30f;
      }
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float pm = fmaxf(fmaxf(s_acc[0][r], s_acc[1][r]),
                       fmaxf(s_acc[2][r], s_acc[3][r]));
      pm = group_max(pm);
      float mn = fmaxf(m[r], pm);
      alpha[r] = __expf(m[r] - mn);
      m[r] = mn;
    }
    float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int f = 0; f < 4; ++f) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = __expf(s_acc[f][r] - m[r]);
        s_acc[f][r] = p;
        psum[r] += p;
      }
    }
    // write P to wave-private LDS in [16 rows][KVB] (swizzled)
#pragma unroll
    for (int f = 0; f < 4; ++f) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int row = (lane >> 4) * 4 + r;
        int cb = (f * 16 + (lane & 15)) * 2;
        *reinterpret_cast<short*>(
            my_p + swz(row, cb, KVB * 2)) = f32_to_bf16(s_acc[f][r]);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      l[r] = l[r] * alpha[r] + group_sum(psum[r]);
#pragma unroll
      for (int f = 0; f < DF; ++f)
        o_acc[f][r] *= alpha[r];
    }

    // ---- O += P @ V: A = P_lds fragments, B = Vt_lds ------------
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {       // KVB/32 k-steps
      bf16x8 pf = *reinterpret_cast<const bf16x8*>(
          my_p + swz(lane & 15, kk * 64 + (lane >> 4) * 16, KVB * 2));
#pragma unroll
      for (int f = 0; f < DF; ++f) {
        bf16x8 vf = *reinterpret_cast<const bf16x8*>(
            v_lds + swz(f * 16 + (lane & 15),
                        kk * 64 + (lane >> 4) * 16, KVB * 2));
        o_acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pf, vf, o_acc[f], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue: O / l ------------------------------------------
  const int orow0 = q0 + (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = orow0 + r;
    if (row >= S) continue;
    float inv = (l[r] > 0.f) ? 1.0f / l[r] : 0.f;
    short* orow = O + qoff + (long)row * D;
#pragma unroll
    for (int f = 0; f < DF; ++f)
      orow[f * 16 + (lane & 15)] = f32_to_bf16(o_acc[f][r] * inv);
  }
}

}  // namespace

torch::Tensor transpose2d(torch::Tensor x);  // gemm_mfma.hip (batched)

torch::Tensor attn_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                       bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(q.dim() == 4, "attn_fwd: [B,H,S,D]");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  TORCH_CHECK(k.sizes() == q.sizes() && v.sizes() == q.sizes(),
              "attn_fwd: expand KV heads before calling (GQA)");
  TORCH_CHECK(D == 64 || D == 128, "attn_fwd: D must be 64 or 128");
  TORCH_CHECK(S % 8 == 0, "attn_fwd: S must be a multiple of 8");
  // pre-transpose V: [B,H,S,D] -> [B,H,D,S] (one memory-bound pass on
  // the batched LDS-tiled transpose; lets the PV B-fragments read
  // 8 bf16 contiguous along kv)
  auto vt = transpose2d(v.reshape({(long)B * H, S, D}))
                .reshape({B, H, D, S});
  auto o = torch::empty_like(q);
  float scale = 1.0f / sqrtf((float)D);
  int grid = B * H * ((S + 63) / 64);
  auto stream = c10::cuda::getCurrentCUDAStream();
#define LAUNCH(DD, CC)                                                   \
  hipLaunchKernelGGL((attn_fwd_kernel<DD, CC>), dim3(grid),              \
                     dim3(NTHREADS), 0, stream.stream(),                 \
                     (const short*)q.data_ptr(),                         \
                     (const short*)k.data_ptr(),                         \
                     (const short*)vt.data_ptr(),                        \
                     (short*)o.data_ptr(), B, H, S, scale)
  if (D == 64) {
    if (causal) { LAUNCH(64, true); } else { LAUNCH(64, false); }
  } else {
    if (causal) { LAUNCH(128, true); } else { LAUNCH(128, false); }
  }
#undef LAUNCH
  HIP_CHECK_LAST();
  return o;
}
