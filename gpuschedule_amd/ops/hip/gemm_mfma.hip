// MFMA bf16 GEMM for gfx950 (MI355X): C[M,N] = A[M,K] . B[N,K]^T (+bias)
//
// Design (CDNA4-first, per the canonical MFMA-GEMM anatomy):
//  * v_mfma_f32_16x16x32_bf16 tiles, fp32 accumulation in AGPRs.
//  * 128x128 block tile, BK=64, 256 threads = 4 waves in a 2x2 grid of
//    64x64 wave tiles (4x4 fragments of 16x16 each).
//  * Both operands staged [out-dim][K] in LDS (the NT layout makes A
//    and B staging symmetric and fully coalesced along K).
//  * XOR bank swizzle (byte ^= (row&7)<<4) on the LDS images so the
//    per-fragment ds_read_b128 (16 lanes reading 16 different rows at
//    one 16-B slot) spreads across 8 slots instead of 8-way
//    conflicting.
//  * Double-buffered LDS; per K-tile: issue next-tile global loads
//    early, MFMA on the current buffer, write the staged registers
//    late (async-stage split, T14), one barrier per tile.
//  * XCD-aware bijective blockIdx swizzle: MI355X has 8 XCDs with
//    private L2s; consecutive blocks share B-panels, so giving each
//    XCD a contiguous grid chunk turns neighbor-tile re-reads into
//    L2 hits.
//  * M/N tails predicated; K must be a multiple of 64 (all shipped
//    model shapes are).
#include <torch/extension.h>
#include <c10/cuda/CUDAStream.h>
#include "common.h"

namespace {

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;

constexpr int BM = 128, BN = 128, BK = 64;
constexpr int NTHREADS = 256;

// LDS row stride in bytes (BK bf16) and the bank swizzle.
//
// g-swizzle (conflict-free for the fragment ds_read_b128 pattern):
// slot' = slot ^ g(row), g(row) = 2*bit1(row) + 4*bit3(row).  A b128
// is serviced in four fixed 16-lane groups that pair {f0-3,f12-15} of
// one k-slot with {f4-11} of the next; with this g every group's 16
// addresses land on the 16 bank-16-units exactly once (checked against
// the gfx950 lane-group table).  The previous (row&7)<<4 XOR left
// 2-way conflicts: rows r and r+8 share both bank parity and XOR.
__device__ __forceinline__ int gsw(int row) {
  return ((row >> 1) & 1) * 2 + ((row >> 3) & 1) * 4;
}
constexpr int ROWB = BK * 2;  // 128 B
__device__ __forceinline__ int swz(int row, int byte_in_row) {
  return row * ROWB + (byte_in_row ^ (gsw(row) << 4));
}

// Tile staging via global_load_lds (direct HBM->LDS DMA, 16 B/lane —
// the compiler never auto-emits it; width 4->16 alone is worth ~1.7x
// on this structure).  The LDS destination is wave-uniform base +
// lane*16 (lane-linear), so the XOR bank swizzle moves to the SOURCE
// address: lane i loads the swizzle-inverse column of its row, making
// the linear LDS image identical to the swizzled image the ds_read
// fragments expect (both-sides-or-neither rule).
// Wave w stages rows {p*32 + w*8 .. +8} of the [128][BK] tile.
// Tail rows are CLAMPED (memory-safe; their products only reach C
// rows/cols that the predicated epilogue never stores).
__device__ __forceinline__ void glds_stage_tile(
    const short* __restrict__ src, long ld, int row0, int rows_max,
    int k0, char* lds_buf, int wid, int lane) {
  const int rl = lane >> 3;                       // row within stripe
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    int rp = p * 32 + wid * 8;
    // g-swizzle inverse on the SOURCE column (LDS row = rp + rl)
    int gcolb = (((lane & 7) ^
                  (((rl >> 1) & 1) * 2 + ((rp >> 3) & 1) * 4)) << 4);
    int gr = row0 + rp + rl;
    if (gr >= rows_max) gr = rows_max - 1;
    const void* g = src + (long)gr * ld + k0 + gcolb / 2;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)
            (lds_buf + rp * ROWB),
        16, 0, 0);
  }
}

__global__ __launch_bounds__(NTHREADS, 2) void gemm_nt_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    int M, int N, int K, int relu) {
  // XCD-aware bijective block swizzle (8 XCDs).
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int nbn = (N + BN - 1) / BN;
  const int bm0 = (wg / nbn) * BM;
  const int bn0 = (wg % nbn) * BN;

  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM * BK * 2];
  // Explicit double-buffer pointers (no runtime-indexed array: rule 20)
  char* Ac = lds;
  char* Bc = lds + BM * ROWB;
  char* An = lds + 2 * BM * ROWB;
  char* Bn = lds + 3 * BM * ROWB;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = (wid >> 1) * 64;    // wave row offset in tile
  const int wc = (wid & 1) * 64;     // wave col offset

  const bool mtail = bm0 + BM > M;
  const bool ntail = bn0 + BN > N;

  float4v acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  glds_stage_tile(A, K, bm0, M, 0, Ac, wid, lane);
  glds_stage_tile(B, K, bn0, N, 0, Bc, wid, lane);
  __syncthreads();  // drains the in-flight LDS-DMA (vmcnt 0) + barrier

  const int nk = K / BK;
  // Fragment LDS byte offsets (row varies per lane: swizzle applies).
  const int arow = wr + (lane & 15);
  const int brow = wc + (lane & 15);
  const int kb = (lane >> 4) * 16;   // 8 bf16 = 16 B per k-subtile slot

  for (int t = 0; t < nk; ++t) {
    if (t + 1 < nk) {
      // issue next tile's LDS-DMA before compute; the end-of-iter
      // __syncthreads() drains it right when the next tile is needed
      glds_stage_tile(A, K, bm0, M, (t + 1) * BK, An, wid, lane);
      glds_stage_tile(B, K, bn0, N, (t + 1) * BK, Bn, wid, lane);
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 af[4], bf[4];
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        af[f] = *reinterpret_cast<const bf16x8*>(
            Ac + swz(arow + f * 16, kk * 64 + kb));
        bf[f] = *reinterpret_cast<const bf16x8*>(
            Bc + swz(brow + f * 16, kk * 64 + kb));
      }
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
    if (t + 1 < nk) {
      char* tmp = Ac; Ac = An; An = tmp;
      tmp = Bc; Bc = Bn; Bn = tmp;
    }
    __syncthreads();
  }

  // Epilogue via LDS bounce: the MFMA C-layout's natural stores are
  // 2-B per lane (32-B transactions, 4x wasted store bandwidth), so
  // each wave writes its fragments into a linear LDS image of the
  // 128x128 tile and the block streams it out in coalesced 16-B rows.
  __syncthreads();
  short* c_lds = (short*)lds;  // [BM][BN] bf16 = 32 KiB
  {
    const int lrow0 = wr + (lane >> 4) * 4;
    const int lcol0 = wc + (lane & 15);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int col = lcol0 + j * 16;
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[i][j][r] + bv;
          if (relu) v = fmaxf(v, 0.f);
          c_lds[(lrow0 + i * 16 + r) * BN + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
  // 128 rows x 256 B: 2048 16-B chunks over 256 threads
#pragma unroll
  for (int p = 0; p < 8; ++p) {
    int idx = p * NTHREADS + tid;
    int row = idx >> 4;
    int colb = (idx & 15) * 8;       // 8 bf16 per chunk
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if ((mtail && grow >= M) || (ntail && gcol >= N)) {
      if (!mtail || grow < M) {
        for (int j = 0; j < 8 && gcol + j < N; ++j)
          C[(long)grow * N + gcol + j] = c_lds[row * BN + colb + j];
      }
      continue;
    }
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BN + colb + j];
    } else {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BN + colb);
    }
  }
}

// ==== 128-tile variant on v_mfma_f32_32x32x16_bf16 ================
// Same staging/LDS images as gemm_nt_kernel; the 32x32 MFMA has a
// ~15% higher throughput ceiling than 16x16 (2382 vs 2075 TF ubench).
// Per wave: 2x2 fragments of 32x32, 16 MFMAs per K-tile.
// C mapping (ISA): col = lane&31, row = (reg&3) + 8*(reg>>2)
//                  + 4*(lane>>5).
__global__ __launch_bounds__(NTHREADS, 2) void gemm_nt32_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    int M, int N, int K, int relu) {
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int nbn = (N + BN - 1) / BN;
  const int bm0 = (wg / nbn) * BM;
  const int bn0 = (wg % nbn) * BN;

  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM * BK * 2];
  char* Ac = lds;
  char* Bc = lds + BM * ROWB;
  char* An = lds + 2 * BM * ROWB;
  char* Bn = lds + 3 * BM * ROWB;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = (wid >> 1) * 64;
  const int wc = (wid & 1) * 64;
  const bool mtail = bm0 + BM > M;
  const bool ntail = bn0 + BN > N;

  float16v acc[2][2];
#pragma unroll
  for (int i = 0; i < 2; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[i][j][r] = 0.f;

  glds_stage_tile(A, K, bm0, M, 0, Ac, wid, lane);
  glds_stage_tile(B, K, bn0, N, 0, Bc, wid, lane);
  __syncthreads();

  const int nk = K / BK;
  const int arow = wr + (lane & 31);
  const int brow = wc + (lane & 31);
  const int kb = (lane >> 5) * 16;   // 8 bf16 within the k=16 subtile

  for (int t = 0; t < nk; ++t) {
    if (t + 1 < nk) {
      glds_stage_tile(A, K, bm0, M, (t + 1) * BK, An, wid, lane);
      glds_stage_tile(B, K, bn0, N, (t + 1) * BK, Bn, wid, lane);
    }
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {   // 4 k-subtiles of 16
      bf16x8 af[2], bf[2];
#pragma unroll
      for (int f = 0; f < 2; ++f) {
        af[f] = *reinterpret_cast<const bf16x8*>(
            Ac + swz(arow + f * 32, kk * 32 + kb));
        bf[f] = *reinterpret_cast<const bf16x8*>(
            Bc + swz(brow + f * 32, kk * 32 + kb));
      }
#pragma unroll
      for (int i = 0; i < 2; ++i)
#pragma unroll
        for (int j = 0; j < 2; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              af[i], bf[j], acc[i][j], 0, 0, 0);
    }
    if (t + 1 < nk) {
      char* tmp = Ac; Ac = An; An = tmp;
      tmp = Bc; Bc = Bn; Bn = tmp;
    }
    __syncthreads();
  }

  __syncthreads();
  short* c_lds = (short*)lds;
  {
#pragma unroll
    for (int i = 0; i < 2; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        int col = wc + j * 32 + (lane & 31);
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int row = wr + i * 32 + (r & 3) + 8 * (r >> 2) +
                    4 * (lane >> 5);
          float v = acc[i][j][r] + bv;
          if (relu) v = fmaxf(v, 0.f);
          c_lds[row * BN + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 8; ++p) {
    int idx = p * NTHREADS + tid;
    int row = idx >> 4;
    int colb = (idx & 15) * 8;
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if (mtail && grow >= M) continue;
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BN + colb + j];
    } else if (!ntail || gcol < N) {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BN + colb);
    }
  }
}

// ==== 256x256-tile phase-interleaved GEMM =========================
// 8 waves (512 threads) in a 2(M) x 4(N) wave grid; per-wave output
// 128x64 = 8x4 fragments; BK=64; both operands' K-tiles double-
// buffered in 128 KiB LDS.  Each K-tile is processed as 4 phases
// (64-row x 32-col C-quadrants, 16 MFMAs each); every phase also
// issues the LDS-DMA for one half-tile of the NEXT K-tile, so 8 glds
// spread across the tile's compute and the end-of-tile barrier waits
// on loads that had ~4 MFMA clusters to land.  s_setprio(1) around
// each MFMA cluster lets the CU scheduler favor compute waves
// (phase-split creates the wave role diversity it needs).
constexpr int BM2 = 256, BN2 = 256, BK2 = 64;
constexpr int NT2 = 512;

__device__ __forceinline__ void glds_stage_half(
    const short* __restrict__ src, long ld, int row0, int rows_max,
    int k0, char* lds_img, int half, int wid, int lane) {
  const int rl = lane >> 3;
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    int rp = half * 128 + p * 64 + wid * 8;
    int gcolb = (((lane & 7) ^
                  (((rl >> 1) & 1) * 2 + ((rp >> 3) & 1) * 4)) << 4);
    int gr = row0 + rp + rl;
    if (gr >= rows_max) gr = rows_max - 1;
    const void* g = src + (long)gr * ld + k0 + gcolb / 2;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)
            (lds_img + rp * ROWB),
        16, 0, 0);
  }
}

__global__ __launch_bounds__(NT2, 1) void gemm_nt256_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    int M, int N, int K, int relu) {
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int nbn = (N + BN2 - 1) / BN2;
  const int bm0 = (wg / nbn) * BM2;
  const int bn0 = (wg % nbn) * BN2;

  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM2 * BK2 * 2];
  char* Ac = lds;
  char* Bc = lds + BM2 * ROWB;
  char* An = lds + 2 * BM2 * ROWB;
  char* Bn = lds + 3 * BM2 * ROWB;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (wid >> 2) * 128;
  const int wc = (wid & 3) * 64;

  float4v acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

#pragma unroll
  for (int h = 0; h < 2; ++h) {
    glds_stage_half(A, K, bm0, M, 0, Ac, h, wid, lane);
    glds_stage_half(B, K, bn0, N, 0, Bc, h, wid, lane);
  }
  __syncthreads();

  const int nk = K / BK2;
  const int arow = (lane & 15);
  const int kb = (lane >> 4) * 16;

  for (int t = 0; t < nk; ++t) {
    const bool pre = t + 1 < nk;
    const int k1 = (t + 1) * BK2;
#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      const int jr = ph >> 1, jc = ph & 1;
      bf16x8 af[4][2], bf[2][2];
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
        for (int fi = 0; fi < 4; ++fi)
          af[fi][kk] = *reinterpret_cast<const bf16x8*>(
              Ac + swz(wr + jr * 64 + fi * 16 + arow, kk * 64 + kb));
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          bf[fj][kk] = *reinterpret_cast<const bf16x8*>(
              Bc + swz(wc + jc * 32 + fj * 16 + arow, kk * 64 + kb));
      }
      if (pre) {
        // one half-tile of the next K-tile per phase: A0 A1 B0 B1
        // (issuing all four at phase 0 measured WORSE at 8192^3:
        // the burst serializes against phase 0's ds_reads)
        if (ph < 2) glds_stage_half(A, K, bm0, M, k1, An, ph, wid, lane);
        else glds_stage_half(B, K, bn0, N, k1, Bn, ph - 2, wid, lane);
      }
      __builtin_amdgcn_s_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int fi = 0; fi < 4; ++fi)
#pragma unroll
          for (int fj = 0; fj < 2; ++fj)
            acc[jr * 4 + fi][jc * 2 + fj] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    af[fi][kk], bf[fj][kk],
                    acc[jr * 4 + fi][jc * 2 + fj], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    if (pre) {
      char* tmp = Ac; Ac = An; An = tmp;
      tmp = Bc; Bc = Bn; Bn = tmp;
    }
    __syncthreads();  // drains this tile's prefetch DMA + barrier
  }

  const bool mtail = bm0 + BM2 > M;
  const bool ntail = bn0 + BN2 > N;
  // LDS-bounce epilogue (same rationale as the 128-tile kernel): the
  // 256x256 bf16 tile exactly fills the 128 KiB LDS.
  __syncthreads();
  short* c_lds = (short*)lds;  // [BM2][BN2]
  {
    const int lrow0 = wr + (lane >> 4) * 4;
    const int lcol0 = wc + (lane & 15);
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int col = lcol0 + j * 16;
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[i][j][r] + bv;
          if (relu) v = fmaxf(v, 0.f);
          c_lds[(lrow0 + i * 16 + r) * BN2 + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 16; ++p) {
    int idx = p * NT2 + (int)threadIdx.x;
    int row = idx >> 5;
    int colb = (idx & 31) * 8;
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if (mtail && grow >= M) continue;
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BN2 + colb + j];
    } else if (!ntail || gcol < N) {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BN2 + colb);
    }
  }
}

// ==== 256-tile, consumption-scheduled staging (counted vmcnt) ======
// A/B RESULT (MI355X, GSAMD_GEMM_CS=1 vs default): wash to -3% at
// 4k/8k square, ~+2% llama-ffn — i.e. the K-tile-boundary vmcnt(0)
// drain this schedule eliminates was NOT a real stall (pieces are >=1
// phase ≈500+ cycles old and HBM latency is already covered).  Kept
// as a gated variant: the counted-wait machinery below is the
// starting point if round 2 raises prefetch distance (3 buffers /
// smaller BK), where counted waits become load-bearing.
//
// The plain 256-tile kernel drains ALL in-flight DMA at each K-tile
// boundary (__syncthreads => s_waitcnt vmcnt(0)), stalling on pieces
// staged as recently as one phase earlier.  Here every wave stages
// EXACTLY the rows it (or its column-sharing partner) consumes, in
// its own consumption order — one A piece + one B piece per phase:
//
//   A slot p: rows  half*128 + (p>>1)*64 + (p&1)*32 + (wid&3)*8
//             (half = wid>>2, the wave's OWN A half; phase p consumes
//              quadrant jr=p>>1 = slots {2jr, 2jr+1})
//   B slot p: rows  (wid&3)*64 + (p>>1)*32 + (p&1)*16 + (wid>>2)*8
//             (the wave's OWN B columns, shared with its wid^4
//              partner; phase p consumes jc=p&1 = slots {2jc, 2jc+1})
//
// With that per-wave FIFO (A0,B0,A1,B1,A2,B2,A3,B3 per tile) the
// phase waits become counted constants instead of full drains:
//   ph0 needs A0,A1,B0,B1 (oldest 4 of the previous tile's 8)
//       -> s_waitcnt vmcnt(4)
//   ph1 needs B2,B3 (forces the rest of the previous tile; the
//       youngest piece is ~2 phase-gaps old by then)
//       -> s_waitcnt vmcnt(2)
//   ph2/ph3 need nothing new -> vmcnt(4)/vmcnt(6) are no-ops.
// Issue happens AFTER the phase barrier, so tile t+2's DMA (into the
// buffer tile t read) cannot start until every wave has left tile t's
// MFMAs.  No __syncthreads inside the K loop at all.
__device__ __forceinline__ void glds_cs_piece(
    const short* __restrict__ src, long ld, int row0, int rows_max,
    int k0, char* lds_img, int rp_base, int lane) {
  const int rl = lane >> 3;
  const int gcolb = (((lane & 7) ^
                      (((rl >> 1) & 1) * 2 +
                       ((rp_base >> 3) & 1) * 4)) << 4);
  int gr = row0 + rp_base + rl;
  if (gr >= rows_max) gr = rows_max - 1;
  const void* g = src + (long)gr * ld + k0 + gcolb / 2;
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)g,
      (__attribute__((address_space(3))) unsigned int*)
          (lds_img + rp_base * ROWB),
      16, 0, 0);
}

__global__ __launch_bounds__(NT2, 1) void gemm_nt256_cs_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    int M, int N, int K, int relu) {
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int nbn = (N + BN2 - 1) / BN2;
  const int bm0 = (wg / nbn) * BM2;
  const int bn0 = (wg % nbn) * BN2;

  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM2 * BK2 * 2];
  char* Ac = lds;
  char* Bc = lds + BM2 * ROWB;
  char* An = lds + 2 * BM2 * ROWB;
  char* Bn = lds + 3 * BM2 * ROWB;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (wid >> 2) * 128;
  const int wc = (wid & 3) * 64;
  // per-wave staged-piece row bases (slot p)
  const int a_rp0 = (wid >> 2) * 128 + (wid & 3) * 8;  // + p-dependent
  const int b_rp0 = (wid & 3) * 64 + (wid >> 2) * 8;

  float4v acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // prologue: tile 0 staged in slot order, full drain once
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    glds_cs_piece(A, K, bm0, M, 0, Ac,
                  a_rp0 + (p >> 1) * 64 + (p & 1) * 32, lane);
    glds_cs_piece(B, K, bn0, N, 0, Bc,
                  b_rp0 + (p >> 1) * 32 + (p & 1) * 16, lane);
  }
  __syncthreads();

  const int nk = K / BK2;
  const int arow = (lane & 15);
  const int kb = (lane >> 4) * 16;

  for (int t = 0; t < nk; ++t) {
    const bool pre = t + 1 < nk;
    const int k1 = (t + 1) * BK2;
#pragma unroll
    for (int ph = 0; ph < 4; ++ph) {
      const int jr = ph >> 1, jc = ph & 1;
      // counted wait for THIS phase's quadrant, then the phase barrier
      // simm16: vmcnt[3:0] | expcnt(7)<<4 | lgkmcnt(15)<<8 (no
      // exp/lgkm wait); vmcnt-only counted waits
      if (ph == 0) __builtin_amdgcn_s_waitcnt(0xf70 | 4);
      else if (ph == 1) __builtin_amdgcn_s_waitcnt(0xf70 | 2);
      // ph2/ph3: everything needed already forced at ph1
      __builtin_amdgcn_s_barrier();
      if (pre) {
        glds_cs_piece(A, K, bm0, M, k1, An,
                      a_rp0 + jr * 64 + jc * 32, lane);
        glds_cs_piece(B, K, bn0, N, k1, Bn,
                      b_rp0 + jr * 32 + jc * 16, lane);
      }
      bf16x8 af[4][2], bf[2][2];
#pragma unroll
      for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
        for (int fi = 0; fi < 4; ++fi)
          af[fi][kk] = *reinterpret_cast<const bf16x8*>(
              Ac + swz(wr + jr * 64 + fi * 16 + arow, kk * 64 + kb));
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          bf[fj][kk] = *reinterpret_cast<const bf16x8*>(
              Bc + swz(wc + jc * 32 + fj * 16 + arow, kk * 64 + kb));
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kk = 0; kk < 2; ++kk)
#pragma unroll
        for (int fi = 0; fi < 4; ++fi)
#pragma unroll
          for (int fj = 0; fj < 2; ++fj)
            acc[jr * 4 + fi][jc * 2 + fj] =
                __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                    af[fi][kk], bf[fj][kk],
                    acc[jr * 4 + fi][jc * 2 + fj], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    if (pre) {
      char* tmp = Ac; Ac = An; An = tmp;
      tmp = Bc; Bc = Bn; Bn = tmp;
    }
  }
  __builtin_amdgcn_s_waitcnt(0xf70);  // drain DMA before LDS reuse
  __builtin_amdgcn_s_barrier();

  const bool mtail = bm0 + BM2 > M;
  const bool ntail = bn0 + BN2 > N;
  __syncthreads();
  short* c_lds = (short*)lds;  // [BM2][BN2]
  {
    const int lrow0 = wr + (lane >> 4) * 4;
    const int lcol0 = wc + (lane & 15);
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int col = lcol0 + j * 16;
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[i][j][r] + bv;
          if (relu) v = fmaxf(v, 0.f);
          c_lds[(lrow0 + i * 16 + r) * BN2 + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 16; ++p) {
    int idx = p * NT2 + (int)threadIdx.x;
    int row = idx >> 5;
    int colb = (idx & 31) * 8;
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if (mtail && grow >= M) continue;
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BN2 + colb + j];
    } else if (!ntail || gcol < N) {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BN2 + colb);
    }
  }
}

// ==== 256x256 / BK=64 consumption-pipelined GEMM (p8) ===============
// Two LDS tile buffers, BK=64, 8 waves (2Mx4N).  Every wave stages,
// per tile, EXACTLY the four 16-row pieces that it (and the waves
// sharing those rows) consume, issued in consumption order during the
// PREVIOUS tile:   ph0: A-jr0   ph1: B-jc0   ph2: B-jc1   ph3: A-jr1.
// Consumption (quadrant phases (jr,jc) = 00,01,10,11):
//   ph0 reads A-jr0 + B-jc0 (the FIFO's oldest 4 instructions),
//   ph1 reads B-jc1, ph2 reads A-jr1, ph3 reads NOTHING (af held from
//   ph2, bf1 held from ph1).
// So every phase's wait is a counted s_waitcnt vmcnt(4) — the DMA
// queue NEVER drains in the main loop (the round-1 kernel's
// __syncthreads drained vmcnt(0) every tile; MfmaUtil 40%).  The wait
// sits BEFORE the phase barrier, so after the barrier every wave's
// needed pieces — including ones staged by its row/column partners,
// who run the identical wait schedule — are in LDS.  One barrier per
// phase is also what makes staging race-free: stage targets buf(t+1),
// whose last reader (tile t-1, which reads nothing at its ph3) is
// provably past its reads once t.ph0's barrier passed.
__device__ __forceinline__ void glds_piece16(
    const short* __restrict__ src, long ld, int row0, int rows_max,
    int k0, char* lds_img, int rp, int lane) {
  const int rl = lane >> 3;
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    int rpp = rp + p * 8;
    int gcolb = (((lane & 7) ^
                  (((rl >> 1) & 1) * 2 + ((rpp >> 3) & 1) * 4)) << 4);
    int gr = row0 + rpp + rl;
    if (gr >= rows_max) gr = rows_max - 1;
    const void* g = src + (long)gr * ld + k0 + gcolb / 2;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)
            (lds_img + rpp * ROWB),
        16, 0, 0);
  }
}

template <bool SPRIO>
__global__ __launch_bounds__(NT2, 1) void gemm_nt256_p8_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    const short* __restrict__ ADD, int M, int N, int K, int relu) {
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  // GROUP_M rasterization: within a band of GM m-blocks, consecutive
  // ids walk the M dimension first, so the ~8 consecutive blocks an
  // XCD chunk runs back-to-back share ONE B column panel (4 MiB = the
  // XCD's L2) instead of each streaming a different one; B's chip-wide
  // re-read factor drops from nbm to nbm/GM.  A panels of the band
  // (GM x 4 MiB) cycle through L3.  (gm, not GM, in the tail band.)
  const int nbn = (N + BN2 - 1) / BN2;
  const int nbm = (M + BM2 - 1) / BM2;
  int bm, bn;
  {
    const int GM = 8;
    int per_band = GM * nbn;
    int band = wg / per_band;
    int rem = wg - band * per_band;
    int gm = min(GM, nbm - band * GM);
    bn = rem / gm;
    bm = band * GM + rem - bn * gm;
  }
  const int bm0 = bm * BM2;
  const int bn0 = bn * BN2;

  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM2 * BK2 * 2];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (wid >> 2) * 128;
  const int wc = (wid & 3) * 64;
  // this wave's staged-piece row bases (16 rows each)
  const int a_rp = (wid >> 2) * 128 + (wid & 3) * 16;  // + jr*64
  const int b_rp = (wid & 3) * 64 + (wid >> 2) * 16;   // + jc*32

  float4v acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK2;
  // prologue: tile 0's pieces in FIFO order (ph0's vmcnt(4) + barrier
  // is the only sync the first tile needs)
  glds_piece16(A, K, bm0, M, 0, lds, a_rp, lane);
  glds_piece16(B, K, bn0, N, 0, lds + BM2 * ROWB, b_rp, lane);
  glds_piece16(B, K, bn0, N, 0, lds + BM2 * ROWB, b_rp + 32, lane);
  glds_piece16(A, K, bm0, M, 0, lds, a_rp + 64, lane);

  const int frow = lane & 15;
  const int kb = (lane >> 4) * 16;

  // SPRIO: static arbitration boost for the second-dispatched half
  // (waves 4-7) instead of per-cluster setprio flips — the younger
  // half is the issue-arbitration loser on every segment.
  if (SPRIO && wid >= 4) __builtin_amdgcn_s_setprio(1);

  for (int t = 0; t < nk; ++t) {
    char* Act = lds + (t & 1) * (2 * BM2 * ROWB);
    char* Bct = Act + BM2 * ROWB;
    char* An = lds + ((t + 1) & 1) * (2 * BM2 * ROWB);
    char* Bn = An + BM2 * ROWB;
    const bool pre = t + 1 < nk;
    const int k1 = (t + 1) * BK2;
    bf16x8 af[4][2], bf0[2][2], bf1[2][2];

    // ---- ph0 (jr0, jc0): needs the FIFO's oldest 4 ----------------
    __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + fi * 16 + frow, kk * 64 + kb));
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        bf0[fj][kk] = *reinterpret_cast<const bf16x8*>(
            Bct + swz(wc + fj * 16 + frow, kk * 64 + kb));
    }
    if (pre) glds_piece16(A, K, bm0, M, k1, An, a_rp, lane);
    if (!SPRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf0[fj][kk], acc[fi][fj], 0, 0, 0);
    if (!SPRIO) __builtin_amdgcn_s_setprio(0);

    // ---- ph1 (jr0, jc1): needs B-jc1 ------------------------------
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    else __builtin_amdgcn_s_waitcnt(0xf70 | 2);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        bf1[fj][kk] = *reinterpret_cast<const bf16x8*>(
            Bct + swz(wc + 32 + fj * 16 + frow, kk * 64 + kb));
    if (pre) glds_piece16(B, K, bn0, N, k1, Bn, b_rp, lane);
    if (!SPRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[fi][2 + fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf1[fj][kk], acc[fi][2 + fj], 0, 0, 0);
    if (!SPRIO) __builtin_amdgcn_s_setprio(0);

    // ---- ph2 (jr1, jc0): needs A-jr1 ------------------------------
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    else __builtin_amdgcn_s_waitcnt(0xf70);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + 64 + fi * 16 + frow, kk * 64 + kb));
    if (pre) glds_piece16(B, K, bn0, N, k1, Bn, b_rp + 32, lane);
    if (!SPRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[4 + fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf0[fj][kk], acc[4 + fi][fj], 0, 0, 0);
    if (!SPRIO) __builtin_amdgcn_s_setprio(0);

    // ---- ph3 (jr1, jc1): everything already in registers ----------
    __builtin_amdgcn_s_barrier();
    if (pre) glds_piece16(A, K, bm0, M, k1, An, a_rp + 64, lane);
    if (!SPRIO) __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[4 + fi][2 + fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf1[fj][kk], acc[4 + fi][2 + fj], 0, 0, 0);
    if (!SPRIO) __builtin_amdgcn_s_setprio(0);
  }

  const bool mtail = bm0 + BM2 > M;
  const bool ntail = bn0 + BN2 > N;
  __builtin_amdgcn_s_waitcnt(0xf70);   // drain before LDS reuse
  __builtin_amdgcn_s_barrier();
  short* c_lds = (short*)lds;  // [BM2][BN2]
  {
    const int lrow0 = wr + (lane >> 4) * 4;
    const int lcol0 = wc + (lane & 15);
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int col = lcol0 + j * 16;
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[i][j][r] + bv;
          if (ADD)  // fused elementwise addend (residual stream)
            v += bf16_to_f32(ADD[(long)min(bm0 + lrow0 + i * 16 + r,
                                           M - 1) * N +
                                 min(bn0 + col, N - 1)]);
          if (relu) v = fmaxf(v, 0.f);
          c_lds[(lrow0 + i * 16 + r) * BN2 + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 16; ++p) {
    int idx = p * NT2 + (int)threadIdx.x;
    int row = idx >> 5;
    int colb = (idx & 31) * 8;
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if (mtail && grow >= M) continue;
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BN2 + colb + j];
    } else if (!ntail || gcol < N) {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BN2 + colb);
    }
  }
}

// ==== 128x128 / BK=64 consumption-pipelined GEMM (p8-128) ==========
// The p8 discipline (per-wave consumption-ordered pieces, counted
// vmcnt(4) at every phase, one barrier per phase, g-swizzle) on a
// 128x128 tile with 4 waves: serves the small-N GEMMs (BERT hidden
// 768) whose grids cannot fill the chip with 256x256 blocks.  64 KiB
// LDS -> 2 blocks/CU, so two blocks' phase schedules interleave on
// each CU.  Replaces the round-1 gemm_nt_kernel (vmcnt(0) drain per
// tile, MfmaUtil 19%) as the small-grid default.
__global__ __launch_bounds__(NTHREADS, 2) void gemm_nt128_p8_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    const short* __restrict__ ADD, int M, int N, int K, int relu) {
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int nbn = (N + BN - 1) / BN;
  const int nbm = (M + BM - 1) / BM;
  int bm, bn;
  {
    const int GM = 8;
    int per_band = GM * nbn;
    int band = wg / per_band;
    int rem = wg - band * per_band;
    int gm = min(GM, nbm - band * GM);
    bn = rem / gm;
    bm = band * GM + rem - bn * gm;
  }
  const int bm0 = bm * BM;
  const int bn0 = bn * BN;

  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM * BK * 2];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (wid >> 1) * 64;    // 2(M) x 2(N) wave grid
  const int wc = (wid & 1) * 64;
  const int a_rp = (wid >> 1) * 64 + (wid & 1) * 16;  // + jr*32
  const int b_rp = (wid & 1) * 64 + (wid >> 1) * 16;  // + jc*32

  if (wid >= 2) __builtin_amdgcn_s_setprio(1);  // younger-half boost

  float4v acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK;
  glds_piece16(A, K, bm0, M, 0, lds, a_rp, lane);
  glds_piece16(B, K, bn0, N, 0, lds + BM * ROWB, b_rp, lane);
  glds_piece16(B, K, bn0, N, 0, lds + BM * ROWB, b_rp + 32, lane);
  glds_piece16(A, K, bm0, M, 0, lds, a_rp + 32, lane);

  const int frow = lane & 15;
  const int kb = (lane >> 4) * 16;

  for (int t = 0; t < nk; ++t) {
    char* Act = lds + (t & 1) * (2 * BM * ROWB);
    char* Bct = Act + BM * ROWB;
    char* An = lds + ((t + 1) & 1) * (2 * BM * ROWB);
    char* Bn = An + BM * ROWB;
    const bool pre = t + 1 < nk;
    const int k1 = (t + 1) * BK;
    bf16x8 af[2][2], bf0[2][2], bf1[2][2];

    // ---- ph0 (jr0, jc0) -------------------------------------------
    __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + fi * 16 + frow, kk * 64 + kb));
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        bf0[fj][kk] = *reinterpret_cast<const bf16x8*>(
            Bct + swz(wc + fj * 16 + frow, kk * 64 + kb));
    }
    if (pre) glds_piece16(A, K, bm0, M, k1, An, a_rp, lane);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf0[fj][kk], acc[fi][fj], 0, 0, 0);

    // ---- ph1 (jr0, jc1) -------------------------------------------
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    else __builtin_amdgcn_s_waitcnt(0xf70 | 2);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        bf1[fj][kk] = *reinterpret_cast<const bf16x8*>(
            Bct + swz(wc + 32 + fj * 16 + frow, kk * 64 + kb));
    if (pre) glds_piece16(B, K, bn0, N, k1, Bn, b_rp, lane);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[fi][2 + fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf1[fj][kk], acc[fi][2 + fj], 0, 0, 0);

    // ---- ph2 (jr1, jc0) -------------------------------------------
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    else __builtin_amdgcn_s_waitcnt(0xf70);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + 32 + fi * 16 + frow, kk * 64 + kb));
    if (pre) glds_piece16(B, K, bn0, N, k1, Bn, b_rp + 32, lane);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[2 + fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf0[fj][kk], acc[2 + fi][fj], 0, 0, 0);

    // ---- ph3 (jr1, jc1) -------------------------------------------
    __builtin_amdgcn_s_barrier();
    if (pre) glds_piece16(A, K, bm0, M, k1, An, a_rp + 32, lane);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[2 + fi][2 + fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf1[fj][kk], acc[2 + fi][2 + fj], 0, 0, 0);
  }

  const bool mtail = bm0 + BM > M;
  const bool ntail = bn0 + BN > N;
  __builtin_amdgcn_s_waitcnt(0xf70);
  __builtin_amdgcn_s_barrier();
  short* c_lds = (short*)lds;  // [BM][BN]
  {
    const int lrow0 = wr + (lane >> 4) * 4;
    const int lcol0 = wc + (lane & 15);
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int col = lcol0 + j * 16;
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[i][j][r] + bv;
          if (ADD)
            v += bf16_to_f32(ADD[(long)min(bm0 + lrow0 + i * 16 + r,
                                           M - 1) * N +
                                 min(bn0 + col, N - 1)]);
          if (relu) v = fmaxf(v, 0.f);
          c_lds[(lrow0 + i * 16 + r) * BN + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 8; ++p) {
    int idx = p * NTHREADS + (int)threadIdx.x;
    int row = idx >> 4;
    int colb = (idx & 15) * 8;
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if (mtail && grow >= M) continue;
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BN + colb + j];
    } else if (!ntail || gcol < N) {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BN + colb);
    }
  }
}

// ==== 256x128 / BK=64 consumption-pipelined GEMM (p8h) ==============
// Half-width p8 for grids where 256x256 blocks cannot fill the chip
// (e.g. Llama-3-8B at batch 1: M = 2048): same consumption-ordered
// counted-vmcnt discipline, 8 waves in a 2(M) x 4(N) grid with
// per-wave 128x32 output, TWO phases per K-tile (jr halves; both fj
// columns in one 16-MFMA cluster).  LDS = 2 x (A 32K + B 16K) = 96 K
// -> 1 block/CU; blocks = (M/256)*(N/128).
__global__ __launch_bounds__(NT2, 1) void gemm_nt256h_p8_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    int M, int N, int K, int relu) {
  constexpr int BNH = 128;
  constexpr int ATILE = 256 * ROWB;      // 32 KiB
  constexpr int BTILE = BNH * ROWB;      // 16 KiB
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int nbn = (N + BNH - 1) / BNH;
  const int nbm = (M + BM2 - 1) / BM2;
  int bm, bn;
  {
    const int GM = 8;
    int per_band = GM * nbn;
    int band = wg / per_band;
    int rem = wg - band * per_band;
    int gm = min(GM, nbm - band * GM);
    bn = rem / gm;
    bm = band * GM + rem - bn * gm;
  }
  const int bm0 = bm * BM2;
  const int bn0 = bn * BNH;

  __shared__ __attribute__((aligned(16))) char lds[2 * (ATILE + BTILE)];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (wid >> 2) * 128;   // 2(M) x 4(N)
  const int wc = (wid & 3) * 32;
  const int a_rp = (wid >> 2) * 128 + (wid & 3) * 16;  // + jr*64
  const int b_rp = wid * 16;                           // B: 128 rows/8

  if (wid >= 4) __builtin_amdgcn_s_setprio(1);

  float4v acc[8][2];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK2;
  // prologue FIFO per tile: [A-jr0(2), B(2), A-jr1(2)]
  glds_piece16(A, K, bm0, M, 0, lds, a_rp, lane);
  glds_piece16(B, K, bn0, N, 0, lds + ATILE, b_rp, lane);
  glds_piece16(A, K, bm0, M, 0, lds, a_rp + 64, lane);

  const int frow = lane & 15;
  const int kb = (lane >> 4) * 16;

  for (int t = 0; t < nk; ++t) {
    char* Act = lds + (t & 1) * (ATILE + BTILE);
    char* Bct = Act + ATILE;
    char* An = lds + ((t + 1) & 1) * (ATILE + BTILE);
    char* Bn = An + ATILE;
    const bool pre = t + 1 < nk;
    const int k1 = (t + 1) * BK2;
    bf16x8 af[4][2], bf[2][2];

    // ---- ph0 (jr0, both fj): needs A-jr0 + B = the oldest 4 --------
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 2);
    else __builtin_amdgcn_s_waitcnt(0xf70 | 2);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + fi * 16 + frow, kk * 64 + kb));
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        bf[fj][kk] = *reinterpret_cast<const bf16x8*>(
            Bct + swz(wc + fj * 16 + frow, kk * 64 + kb));
    }
    if (pre) {
      glds_piece16(A, K, bm0, M, k1, An, a_rp, lane);
      glds_piece16(B, K, bn0, N, k1, Bn, b_rp, lane);
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf[fj][kk], acc[fi][fj], 0, 0, 0);

    // ---- ph1 (jr1): needs A-jr1 ------------------------------------
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    else __builtin_amdgcn_s_waitcnt(0xf70);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + 64 + fi * 16 + frow, kk * 64 + kb));
    if (pre) glds_piece16(A, K, bm0, M, k1, An, a_rp + 64, lane);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[4 + fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf[fj][kk], acc[4 + fi][fj], 0, 0, 0);
  }

  const bool mtail = bm0 + BM2 > M;
  const bool ntail = bn0 + BNH > N;
  __builtin_amdgcn_s_waitcnt(0xf70);
  __builtin_amdgcn_s_barrier();
  short* c_lds = (short*)lds;  // [256][BNH] = 64 KiB
  {
    const int lrow0 = wr + (lane >> 4) * 4;
    const int lcol0 = wc + (lane & 15);
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        int col = lcol0 + j * 16;
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[i][j][r] + bv;
          if (relu) v = fmaxf(v, 0.f);
          c_lds[(lrow0 + i * 16 + r) * BNH + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 8; ++p) {
    int idx = p * NT2 + (int)threadIdx.x;
    int row = idx >> 4;
    int colb = (idx & 15) * 8;
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if (mtail && grow >= M) continue;
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BNH + colb + j];
    } else if (!ntail || gcol < N) {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BNH + colb);
    }
  }
}

// ==== NN GEMM: C[M,N] = A[M,R] . B[R,N] (B stored red-major) ========
// The backward data GEMM dX = dY . W consumes W exactly as stored, so
// this kernel removes the explicit W^T transpose pass (round 1:
// transpose kernels = 5.3% of the Llama-8B job).  A side: identical
// to p8 (consumption-staged counted-vmcnt pipeline, g-swizzle).
// B side: each K-tile's [64 red][256 out] slab is stored as 128-B
// subtiles [4 red][16 out] and fragments are read with
// ds_read_b64_tr_b16 — gfx950's hardware 4x4-bf16 transpose.
// Semantics (probed on hardware, scripts/tr16_probe.py): each 16-lane
// group gathers 16 lane-addressed 8-B chunks as a [4 red][16 out]
// tile — lane 16g+j ADDRESSES row j>>2, chunk j&3 and RECEIVES column
// j (4 red values of out-col j).  Subtile order perm(r4) (swap bits
// 0<->1 within each group of 4; an involution) makes the two tiles of
// each 32-lane bank half 256 B contiguous -> conflict-free.
__device__ __forceinline__ int perm_nn(int r4) {
  return (r4 & ~3) | ((r4 & 1) << 1) | ((r4 >> 1) & 1);
}

// Stage one 16-out-col band (all 64 red rows = 16 subtiles = 2 KiB)
// of the B slab: 2 glds instrs of 8 subtiles.  tile red rows must be
// full (red tails are the caller's padding job: red is the REDUCTION
// dim, clamped rows would corrupt the product).
__device__ __forceinline__ void glds_nn_piece(
    const short* __restrict__ src, long ld, int red0, int c16,
    int tile_out0, int outs_max, char* img, int lane) {
  const int red_in = (lane & 7) >> 1;
  const int out8 = (lane & 1) * 8;
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    const int pidx = p * 8 + (lane >> 3);
    const int r4 = perm_nn(pidx);          // involution
    const int red = red0 + r4 * 4 + red_in;
    int out = tile_out0 + c16 * 16 + out8;
    if (out + 7 >= outs_max) out = max(outs_max - 8, 0);
    const void* g = src + (long)red * ld + out;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)
            (img + (c16 * 16 + p * 8) * 128 + (lane & 7) * 16),
        16, 0, 0);
  }
}

// B fragment (16 out cols at fc0, red = kk*32 + (lane>>4)*8 + 0..7)
// via two hardware-transpose reads.  NO wait here: the caller issues
// every fragment's reads, then one s_waitcnt lgkmcnt(0).
__device__ __forceinline__ void bfrag_nn_issue(
    const char* img, int fc0, int kk, int lane,
    short4v* lo, short4v* hi) {
  const int g = lane >> 4;
  const int j = lane & 15;
  const int c16 = fc0 >> 4;
  const int r4a = kk * 8 + g * 2;
  const int woff = (j >> 2) * 32 + (j & 3) * 8;
  unsigned a0 = (unsigned)(size_t)(img) +
                (unsigned)((c16 * 16 + perm_nn(r4a)) * 128 + woff);
  unsigned a1 = (unsigned)(size_t)(img) +
                (unsigned)((c16 * 16 + perm_nn(r4a + 1)) * 128 + woff);
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0"
               : "=v"(*lo) : "v"(a0));
  asm volatile("ds_read_b64_tr_b16 %0, %1 offset:0"
               : "=v"(*hi) : "v"(a1));
}

__device__ __forceinline__ bf16x8 bfrag_nn_pack(short4v lo, short4v hi) {
  short8 s8 = __builtin_shufflevector(lo, hi, 0, 1, 2, 3, 4, 5, 6, 7);
  return *reinterpret_cast<bf16x8*>(&s8);
}

__global__ __launch_bounds__(NT2, 1) void gemm_nn_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    int M, int N, int R, int relu) {
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int nbn = (N + BN2 - 1) / BN2;
  const int nbm = (M + BM2 - 1) / BM2;
  int bm, bn;
  {
    const int GM = 8;
    int per_band = GM * nbn;
    int band = wg / per_band;
    int rem = wg - band * per_band;
    int gm = min(GM, nbm - band * GM);
    bn = rem / gm;
    bm = band * GM + rem - bn * gm;
  }
  const int bm0 = bm * BM2;
  const int bn0 = bn * BN2;

  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM2 * BK2 * 2];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (wid >> 2) * 128;
  const int wc = (wid & 3) * 64;
  const int a_rp = (wid >> 2) * 128 + (wid & 3) * 16;  // + jr*64
  const int b_c16 = (wid & 3) * 4 + (wid >> 2);        // + jc*2

  if (true && wid >= 4) __builtin_amdgcn_s_setprio(1);

  float4v acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nk = R / BK2;
  glds_piece16(A, R, bm0, M, 0, lds, a_rp, lane);
  glds_nn_piece(B, N, 0, b_c16, bn0, N, lds + BM2 * ROWB, lane);
  glds_nn_piece(B, N, 0, b_c16 + 2, bn0, N, lds + BM2 * ROWB, lane);
  glds_piece16(A, R, bm0, M, 0, lds, a_rp + 64, lane);

  const int frow = lane & 15;
  const int kb = (lane >> 4) * 16;

  for (int t = 0; t < nk; ++t) {
    char* Act = lds + (t & 1) * (2 * BM2 * ROWB);
    char* Bct = Act + BM2 * ROWB;
    char* An = lds + ((t + 1) & 1) * (2 * BM2 * ROWB);
    char* Bn = An + BM2 * ROWB;
    const bool pre = t + 1 < nk;
    const int k1 = (t + 1) * BK2;
    bf16x8 af[4][2], bf0[2][2], bf1[2][2];
    short4v tlo[2][2], thi[2][2];

    // ---- ph0 (jr0, jc0) -------------------------------------------
    __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + fi * 16 + frow, kk * 64 + kb));
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        bfrag_nn_issue(Bct, wc + fj * 16, kk, lane,
                       &tlo[fj][kk], &thi[fj][kk]);
    }
    asm volatile("s_waitcnt lgkmcnt(0)");
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        bf0[fj][kk] = bfrag_nn_pack(tlo[fj][kk], thi[fj][kk]);
    if (pre) glds_piece16(A, R, bm0, M, k1, An, a_rp, lane);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf0[fj][kk], acc[fi][fj], 0, 0, 0);

    // ---- ph1 (jr0, jc1) -------------------------------------------
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    else __builtin_amdgcn_s_waitcnt(0xf70 | 2);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        bfrag_nn_issue(Bct, wc + 32 + fj * 16, kk, lane,
                       &tlo[fj][kk], &thi[fj][kk]);
    asm volatile("s_waitcnt lgkmcnt(0)");
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        bf1[fj][kk] = bfrag_nn_pack(tlo[fj][kk], thi[fj][kk]);
    if (pre) glds_nn_piece(B, N, k1, b_c16, bn0, N, Bn, lane);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[fi][2 + fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf1[fj][kk], acc[fi][2 + fj], 0, 0, 0);

    // ---- ph2 (jr1, jc0) -------------------------------------------
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    else __builtin_amdgcn_s_waitcnt(0xf70);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + 64 + fi * 16 + frow, kk * 64 + kb));
    if (pre) glds_nn_piece(B, N, k1, b_c16 + 2, bn0, N, Bn, lane);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[4 + fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf0[fj][kk], acc[4 + fi][fj], 0, 0, 0);

    // ---- ph3 (jr1, jc1) -------------------------------------------
    __builtin_amdgcn_s_barrier();
    if (pre) glds_piece16(A, R, bm0, M, k1, An, a_rp + 64, lane);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[4 + fi][2 + fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf1[fj][kk], acc[4 + fi][2 + fj], 0, 0, 0);
  }

  const bool mtail = bm0 + BM2 > M;
  const bool ntail = bn0 + BN2 > N;
  __builtin_amdgcn_s_waitcnt(0xf70);
  __builtin_amdgcn_s_barrier();
  short* c_lds = (short*)lds;  // [BM2][BN2]
  {
    const int lrow0 = wr + (lane >> 4) * 4;
    const int lcol0 = wc + (lane & 15);
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int col = lcol0 + j * 16;
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[i][j][r] + bv;
          if (relu) v = fmaxf(v, 0.f);
          c_lds[(lrow0 + i * 16 + r) * BN2 + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 16; ++p) {
    int idx = p * NT2 + (int)threadIdx.x;
    int row = idx >> 5;
    int colb = (idx & 31) * 8;
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if (mtail && grow >= M) continue;
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BN2 + colb + j];
    } else if (!ntail || gcol < N) {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BN2 + colb);
    }
  }
}

// ==== p12: p8 on v_mfma_f32_32x32x16_bf16 ==========================
// Same consumption-staged counted-vmcnt schedule as p8, but the MFMA
// is 32x32x16 (back-to-back issue 32 cyc/SIMD for 16384 MACs vs
// 17 cyc for 16x16x32's 8192 — ~6% fewer issue cycles and a 2495 vs
// 2075 TF ubench ceiling).  The 32-row fragment pattern needs its own
// conflict-free LDS permutation: a b128 lane group reads 16 DISTINCT
// rows of {0-3,12-15,20-27} (+32-offset) at ONE k-slot, so
// slot' = slot ^ ((row>>1)&7) makes each group's 16 addresses hit the
// 16 bank-16-units exactly once (checked against the lane-group
// table; the p8 g-swizzle collides here: rows 0 and 20 alias).
__device__ __forceinline__ int swz32(int row, int byte_in_row) {
  return row * ROWB + (byte_in_row ^ (((row >> 1) & 7) << 4));
}

__device__ __forceinline__ void glds_piece16_32(
    const short* __restrict__ src, long ld, int row0, int rows_max,
    int k0, char* lds_img, int rp, int lane) {
  const int rl = lane >> 3;
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    int rpp = rp + p * 8;
    int gcolb = (((lane & 7) ^
                  ((((rpp >> 3) & 1) * 4 + (rl >> 1)) & 7)) << 4);
    int gr = row0 + rpp + rl;
    if (gr >= rows_max) gr = rows_max - 1;
    const void* g = src + (long)gr * ld + k0 + gcolb / 2;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)
            (lds_img + rpp * ROWB),
        16, 0, 0);
  }
}

__global__ __launch_bounds__(NT2, 1) void gemm_nt256_p12_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    int M, int N, int K, int relu) {
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int nbn = (N + BN2 - 1) / BN2;
  const int nbm = (M + BM2 - 1) / BM2;
  int bm, bn;
  {
    const int GM = 8;
    int per_band = GM * nbn;
    int band = wg / per_band;
    int rem = wg - band * per_band;
    int gm = min(GM, nbm - band * GM);
    bn = rem / gm;
    bm = band * GM + rem - bn * gm;
  }
  const int bm0 = bm * BM2;
  const int bn0 = bn * BN2;

  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM2 * BK2 * 2];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (wid >> 2) * 128;
  const int wc = (wid & 3) * 64;
  const int a_rp = (wid >> 2) * 128 + (wid & 3) * 16;  // + jr*64
  const int b_rp = (wid & 3) * 64 + (wid >> 2) * 16;   // + jc*32

  float16v acc[4][2];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 2; ++j)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[i][j][r] = 0.f;

  const int nk = K / BK2;
  glds_piece16_32(A, K, bm0, M, 0, lds, a_rp, lane);
  glds_piece16_32(B, K, bn0, N, 0, lds + BM2 * ROWB, b_rp, lane);
  glds_piece16_32(B, K, bn0, N, 0, lds + BM2 * ROWB, b_rp + 32, lane);
  glds_piece16_32(A, K, bm0, M, 0, lds, a_rp + 64, lane);

  const int frow = lane & 31;          // fragment row within 32
  const int kb = (lane >> 5) * 16;     // k-half slot (16 B)

  for (int t = 0; t < nk; ++t) {
    char* Act = lds + (t & 1) * (2 * BM2 * ROWB);
    char* Bct = Act + BM2 * ROWB;
    char* An = lds + ((t + 1) & 1) * (2 * BM2 * ROWB);
    char* Bn = An + BM2 * ROWB;
    const bool pre = t + 1 < nk;
    const int k1 = (t + 1) * BK2;
    bf16x8 af[2][4], bf0[4], bf1[4];

    // ---- ph0 (jr0, jc0) -------------------------------------------
    __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz32(wr + fi * 32 + frow, kk * 32 + kb));
      bf0[kk] = *reinterpret_cast<const bf16x8*>(
          Bct + swz32(wc + frow, kk * 32 + kb));
    }
    if (pre) glds_piece16_32(A, K, bm0, M, k1, An, a_rp, lane);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
        acc[fi][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            af[fi][kk], bf0[kk], acc[fi][0], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- ph1 (jr0, jc1) -------------------------------------------
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    else __builtin_amdgcn_s_waitcnt(0xf70 | 2);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
      bf1[kk] = *reinterpret_cast<const bf16x8*>(
          Bct + swz32(wc + 32 + frow, kk * 32 + kb));
    if (pre) glds_piece16_32(B, K, bn0, N, k1, Bn, b_rp, lane);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
        acc[fi][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            af[fi][kk], bf1[kk], acc[fi][1], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- ph2 (jr1, jc0) -------------------------------------------
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    else __builtin_amdgcn_s_waitcnt(0xf70);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz32(wr + 64 + fi * 32 + frow, kk * 32 + kb));
    if (pre) glds_piece16_32(B, K, bn0, N, k1, Bn, b_rp + 32, lane);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
        acc[2 + fi][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            af[fi][kk], bf0[kk], acc[2 + fi][0], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- ph3 (jr1, jc1) -------------------------------------------
    __builtin_amdgcn_s_barrier();
    if (pre) glds_piece16_32(A, K, bm0, M, k1, An, a_rp + 64, lane);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 4; ++kk)
#pragma unroll
      for (int fi = 0; fi < 2; ++fi)
        acc[2 + fi][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
            af[fi][kk], bf1[kk], acc[2 + fi][1], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  }

  const bool mtail = bm0 + BM2 > M;
  const bool ntail = bn0 + BN2 > N;
  __builtin_amdgcn_s_waitcnt(0xf70);
  __builtin_amdgcn_s_barrier();
  short* c_lds = (short*)lds;  // [BM2][BN2]
  {
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        int col = wc + j * 32 + (lane & 31);
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int row = wr + i * 32 + (r & 3) + 8 * (r >> 2) +
                    4 * (lane >> 5);
          float v = acc[i][j][r] + bv;
          if (relu) v = fmaxf(v, 0.f);
          c_lds[row * BN2 + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 16; ++p) {
    int idx = p * NT2 + (int)threadIdx.x;
    int row = idx >> 5;
    int colb = (idx & 31) * 8;
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if (mtail && grow >= M) continue;
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BN2 + colb + j];
    } else if (!ntail || gcol < N) {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BN2 + colb);
    }
  }
}

// ==== p9: p8 with region-recycled deep staging ======================
// Same consumption-ordered pieces as p8, but every piece of tile u is
// issued ~4 phases (one full tile) before its read instead of 1-3:
// each buffer REGION is recycled immediately after its single read
// (A-jr0 of buf t is only read at t.ph0, so t+2's A-jr0 can be staged
// into it at t.ph1 — the phase barrier proves all waves are past the
// read).  Every steady-state wait becomes vmcnt(10) — five pieces
// younger than the needed one stay in flight — so the waits
// essentially never block.
//   stage(u).A-jr0 @ (u-2).ph1     stage(u).B-jc0 @ (u-2).ph2
//   stage(u).B-jc1 @ (u-2).ph3     stage(u).A-jr1 @ (u-1).ph0
__global__ __launch_bounds__(NT2, 1) void gemm_nt256_p9_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    int M, int N, int K, int relu) {
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int nbn = (N + BN2 - 1) / BN2;
  const int bm0 = (wg / nbn) * BM2;
  const int bn0 = (wg % nbn) * BN2;

  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM2 * BK2 * 2];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (wid >> 2) * 128;
  const int wc = (wid & 3) * 64;
  // this wave's staged-piece row bases (16 rows each)
  const int a_rp = (wid >> 2) * 128 + (wid & 3) * 16;  // + jr*64
  const int b_rp = (wid & 3) * 64 + (wid >> 2) * 16;   // + jc*32

  float4v acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK2;
  // prologue: tile 0's pieces in FIFO order (ph0's vmcnt(4) + barrier
  // is the only sync the first tile needs)
  glds_piece16(A, K, bm0, M, 0, lds, a_rp, lane);
  glds_piece16(B, K, bn0, N, 0, lds + BM2 * ROWB, b_rp, lane);
  glds_piece16(B, K, bn0, N, 0, lds + BM2 * ROWB, b_rp + 32, lane);
  glds_piece16(A, K, bm0, M, 0, lds, a_rp + 64, lane);
  if (nk > 1) {
    char* A1 = lds + 2 * BM2 * ROWB;
    glds_piece16(A, K, bm0, M, BK2, A1, a_rp, lane);
    glds_piece16(B, K, bn0, N, BK2, A1 + BM2 * ROWB, b_rp, lane);
    glds_piece16(B, K, bn0, N, BK2, A1 + BM2 * ROWB, b_rp + 32, lane);
  }

  const int frow = lane & 15;
  const int kb = (lane >> 4) * 16;

  for (int t = 0; t < nk; ++t) {
    char* Act = lds + (t & 1) * (2 * BM2 * ROWB);
    char* Bct = Act + BM2 * ROWB;
    char* An = lds + ((t + 1) & 1) * (2 * BM2 * ROWB);
    const bool pre = t + 1 < nk;
    const bool pre2 = t + 2 < nk;
    const int k1 = (t + 1) * BK2;
    const int k2 = (t + 2) * BK2;
    bf16x8 af[4][2], bf0[2][2], bf1[2][2];

    // ---- ph0 (jr0, jc0): needs the FIFO's oldest 4 ----------------
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 10);
    else __builtin_amdgcn_s_waitcnt(0xf70 | 4);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + fi * 16 + frow, kk * 64 + kb));
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        bf0[fj][kk] = *reinterpret_cast<const bf16x8*>(
            Bct + swz(wc + fj * 16 + frow, kk * 64 + kb));
    }
    if (pre) glds_piece16(A, K, bm0, M, k1, An, a_rp + 64, lane);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf0[fj][kk], acc[fi][fj], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- ph1 (jr0, jc1): needs B-jc1 ------------------------------
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 10);
    else __builtin_amdgcn_s_waitcnt(0xf70 | 2);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        bf1[fj][kk] = *reinterpret_cast<const bf16x8*>(
            Bct + swz(wc + 32 + fj * 16 + frow, kk * 64 + kb));
    if (pre2) glds_piece16(A, K, bm0, M, k2, Act, a_rp, lane);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[fi][2 + fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf1[fj][kk], acc[fi][2 + fj], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- ph2 (jr1, jc0): needs A-jr1 ------------------------------
    if (pre2) __builtin_amdgcn_s_waitcnt(0xf70 | 10);
    else if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 8);
    else __builtin_amdgcn_s_waitcnt(0xf70);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + 64 + fi * 16 + frow, kk * 64 + kb));
    if (pre2) glds_piece16(B, K, bn0, N, k2, Bct, b_rp, lane);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[4 + fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf0[fj][kk], acc[4 + fi][fj], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- ph3 (jr1, jc1): everything already in registers ----------
    __builtin_amdgcn_s_barrier();
    if (pre2) glds_piece16(B, K, bn0, N, k2, Bct, b_rp + 32, lane);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[4 + fi][2 + fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf1[fj][kk], acc[4 + fi][2 + fj], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  }

  const bool mtail = bm0 + BM2 > M;
  const bool ntail = bn0 + BN2 > N;
  __builtin_amdgcn_s_waitcnt(0xf70);   // drain before LDS reuse
  __builtin_amdgcn_s_barrier();
  short* c_lds = (short*)lds;  // [BM2][BN2]
  {
    const int lrow0 = wr + (lane >> 4) * 4;
    const int lcol0 = wc + (lane & 15);
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int col = lcol0 + j * 16;
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[i][j][r] + bv;
          if (relu) v = fmaxf(v, 0.f);
          c_lds[(lrow0 + i * 16 + r) * BN2 + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 16; ++p) {
    int idx = p * NT2 + (int)threadIdx.x;
    int row = idx >> 5;
    int colb = (idx & 31) * 8;
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if (mtail && grow >= M) continue;
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BN2 + colb + j];
    } else if (!ntail || gcol < N) {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BN2 + colb);
    }
  }
}


// ==== p11: p9 merged to TWO megaphases per tile (one vmcnt wait +
// two barriers per tile; 32-MFMA clusters) =========================
__global__ __launch_bounds__(NT2, 1) void gemm_nt256_p11_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    int M, int N, int K, int relu) {
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int nbn = (N + BN2 - 1) / BN2;
  const int bm0 = (wg / nbn) * BM2;
  const int bn0 = (wg % nbn) * BN2;

  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM2 * BK2 * 2];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (wid >> 2) * 128;
  const int wc = (wid & 3) * 64;
  // this wave's staged-piece row bases (16 rows each)
  const int a_rp = (wid >> 2) * 128 + (wid & 3) * 16;  // + jr*64
  const int b_rp = (wid & 3) * 64 + (wid >> 2) * 16;   // + jc*32

  float4v acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK2;
  // prologue: tile 0's pieces in FIFO order (ph0's vmcnt(4) + barrier
  // is the only sync the first tile needs)
  glds_piece16(A, K, bm0, M, 0, lds, a_rp, lane);
  glds_piece16(B, K, bn0, N, 0, lds + BM2 * ROWB, b_rp, lane);
  glds_piece16(B, K, bn0, N, 0, lds + BM2 * ROWB, b_rp + 32, lane);
  glds_piece16(A, K, bm0, M, 0, lds, a_rp + 64, lane);
  if (nk > 1) {
    char* A1 = lds + 2 * BM2 * ROWB;
    glds_piece16(A, K, bm0, M, BK2, A1, a_rp, lane);
    glds_piece16(B, K, bn0, N, BK2, A1 + BM2 * ROWB, b_rp, lane);
    glds_piece16(B, K, bn0, N, BK2, A1 + BM2 * ROWB, b_rp + 32, lane);
  }

  const int frow = lane & 15;
  const int kb = (lane >> 4) * 16;

  for (int t = 0; t < nk; ++t) {
    char* Act = lds + (t & 1) * (2 * BM2 * ROWB);
    char* Bct = Act + BM2 * ROWB;
    char* An = lds + ((t + 1) & 1) * (2 * BM2 * ROWB);
    const bool pre = t + 1 < nk;
    const bool pre2 = t + 2 < nk;
    const int k1 = (t + 1) * BK2;
    const int k2 = (t + 2) * BK2;
    bf16x8 af[4][2], bf0[2][2], bf1[2][2];

    // ---- megaphase 0 (jr0 x both jc): one deep wait proves ALL of
    // tile t landed (each piece is >= 3 megaphases old); 16 fragment
    // reads; 32 MFMAs in one cluster.
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 6);
    else __builtin_amdgcn_s_waitcnt(0xf70);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + fi * 16 + frow, kk * 64 + kb));
#pragma unroll
      for (int fj = 0; fj < 2; ++fj) {
        bf0[fj][kk] = *reinterpret_cast<const bf16x8*>(
            Bct + swz(wc + fj * 16 + frow, kk * 64 + kb));
        bf1[fj][kk] = *reinterpret_cast<const bf16x8*>(
            Bct + swz(wc + 32 + fj * 16 + frow, kk * 64 + kb));
      }
    }
    if (pre) glds_piece16(A, K, bm0, M, k1, An, a_rp + 64, lane);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj) {
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf0[fj][kk], acc[fi][fj], 0, 0, 0);
          acc[fi][2 + fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf1[fj][kk], acc[fi][2 + fj], 0, 0, 0);
        }
    __builtin_amdgcn_s_setprio(0);

    // ---- megaphase 1 (jr1 x both jc): every region of tile t's
    // buffers has been read, so all three of tile t+2's early pieces
    // stage here (into the just-freed regions); af reloads in place.
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + 64 + fi * 16 + frow, kk * 64 + kb));
    if (pre2) {
      glds_piece16(A, K, bm0, M, k2, Act, a_rp, lane);
      glds_piece16(B, K, bn0, N, k2, Bct, b_rp, lane);
      glds_piece16(B, K, bn0, N, k2, Bct, b_rp + 32, lane);
    }
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj) {
          acc[4 + fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf0[fj][kk], acc[4 + fi][fj], 0, 0, 0);
          acc[4 + fi][2 + fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf1[fj][kk], acc[4 + fi][2 + fj], 0, 0, 0);
        }
    __builtin_amdgcn_s_setprio(0);
  }

  const bool mtail = bm0 + BM2 > M;
  const bool ntail = bn0 + BN2 > N;
  __builtin_amdgcn_s_waitcnt(0xf70);   // drain before LDS reuse
  __builtin_amdgcn_s_barrier();
  short* c_lds = (short*)lds;  // [BM2][BN2]
  {
    const int lrow0 = wr + (lane >> 4) * 4;
    const int lcol0 = wc + (lane & 15);
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int col = lcol0 + j * 16;
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[i][j][r] + bv;
          if (relu) v = fmaxf(v, 0.f);
          c_lds[(lrow0 + i * 16 + r) * BN2 + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 16; ++p) {
    int idx = p * NT2 + (int)threadIdx.x;
    int row = idx >> 5;
    int colb = (idx & 31) * 8;
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if (mtail && grow >= M) continue;
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BN2 + colb + j];
    } else if (!ntail || gcol < N) {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BN2 + colb);
    }
  }
}



// ==== p10: p9 + fragment reads pipelined into the previous
// phase's MFMA cluster (single deep wait per tile) ================
__global__ __launch_bounds__(NT2, 2) void gemm_nt256_p10_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    int M, int N, int K, int relu) {
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int nbn = (N + BN2 - 1) / BN2;
  const int bm0 = (wg / nbn) * BM2;
  const int bn0 = (wg % nbn) * BN2;

  __shared__ __attribute__((aligned(16))) char lds[2 * 2 * BM2 * BK2 * 2];

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (wid >> 2) * 128;
  const int wc = (wid & 3) * 64;
  // this wave's staged-piece row bases (16 rows each)
  const int a_rp = (wid >> 2) * 128 + (wid & 3) * 16;  // + jr*64
  const int b_rp = (wid & 3) * 64 + (wid >> 2) * 16;   // + jc*32

  float4v acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK2;
  // prologue: tile 0's pieces in FIFO order (ph0's vmcnt(4) + barrier
  // is the only sync the first tile needs)
  glds_piece16(A, K, bm0, M, 0, lds, a_rp, lane);
  glds_piece16(B, K, bn0, N, 0, lds + BM2 * ROWB, b_rp, lane);
  glds_piece16(B, K, bn0, N, 0, lds + BM2 * ROWB, b_rp + 32, lane);
  glds_piece16(A, K, bm0, M, 0, lds, a_rp + 64, lane);
  if (nk > 1) {
    char* A1 = lds + 2 * BM2 * ROWB;
    glds_piece16(A, K, bm0, M, BK2, A1, a_rp, lane);
    glds_piece16(B, K, bn0, N, BK2, A1 + BM2 * ROWB, b_rp, lane);
    glds_piece16(B, K, bn0, N, BK2, A1 + BM2 * ROWB, b_rp + 32, lane);
  }

  const int frow = lane & 15;
  const int kb = (lane >> 4) * 16;

  for (int t = 0; t < nk; ++t) {
    char* Act = lds + (t & 1) * (2 * BM2 * ROWB);
    char* Bct = Act + BM2 * ROWB;
    char* An = lds + ((t + 1) & 1) * (2 * BM2 * ROWB);
    const bool pre = t + 1 < nk;
    const bool pre2 = t + 2 < nk;
    const int k1 = (t + 1) * BK2;
    const int k2 = (t + 2) * BK2;
    bf16x8 af[4][2], bf0[2][2], bf1[2][2];

    // ---- ph0: ONE deep wait proves ALL of tile t landed (pieces are
    // a full tile old); later phases then read EARLY — each phase's
    // fragments are loaded during the PREVIOUS phase's MFMA cluster,
    // so the reads' LDS latency hides under matrix work.
    if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 6);
    else __builtin_amdgcn_s_waitcnt(0xf70);
    __builtin_amdgcn_s_barrier();
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
        af[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + fi * 16 + frow, kk * 64 + kb));
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)
        bf0[fj][kk] = *reinterpret_cast<const bf16x8*>(
            Bct + swz(wc + fj * 16 + frow, kk * 64 + kb));
    }
    if (pre) glds_piece16(A, K, bm0, M, k1, An, a_rp + 64, lane);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf0[fj][kk], acc[fi][fj], 0, 0, 0);
#pragma unroll
      for (int fj = 0; fj < 2; ++fj)   // ph1's fragments, early
        bf1[fj][kk] = *reinterpret_cast<const bf16x8*>(
            Bct + swz(wc + 32 + fj * 16 + frow, kk * 64 + kb));
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- ph1 (jr0, jc1): bf1 already read during ph0 ---------------
    __builtin_amdgcn_s_barrier();
    if (pre2) glds_piece16(A, K, bm0, M, k2, Act, a_rp, lane);
    bf16x8 af2[4][2];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[fi][2 + fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi][kk], bf1[fj][kk], acc[fi][2 + fj], 0, 0, 0);
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)   // ph2's fragments, early
        af2[fi][kk] = *reinterpret_cast<const bf16x8*>(
            Act + swz(wr + 64 + fi * 16 + frow, kk * 64 + kb));
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- ph2 (jr1, jc0): af2 already read during ph1 ---------------
    __builtin_amdgcn_s_barrier();
    if (pre2) glds_piece16(B, K, bn0, N, k2, Bct, b_rp, lane);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[4 + fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af2[fi][kk], bf0[fj][kk], acc[4 + fi][fj], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);

    // ---- ph3 (jr1, jc1): everything already in registers ----------
    __builtin_amdgcn_s_barrier();
    if (pre2) glds_piece16(B, K, bn0, N, k2, Bct, b_rp + 32, lane);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int kk = 0; kk < 2; ++kk)
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 2; ++fj)
          acc[4 + fi][2 + fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af2[fi][kk], bf1[fj][kk], acc[4 + fi][2 + fj], 0, 0, 0);
    __builtin_amdgcn_s_setprio(0);
  }

  const bool mtail = bm0 + BM2 > M;
  const bool ntail = bn0 + BN2 > N;
  __builtin_amdgcn_s_waitcnt(0xf70);   // drain before LDS reuse
  __builtin_amdgcn_s_barrier();
  short* c_lds = (short*)lds;  // [BM2][BN2]
  {
    const int lrow0 = wr + (lane >> 4) * 4;
    const int lcol0 = wc + (lane & 15);
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int col = lcol0 + j * 16;
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[i][j][r] + bv;
          if (relu) v = fmaxf(v, 0.f);
          c_lds[(lrow0 + i * 16 + r) * BN2 + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 16; ++p) {
    int idx = p * NT2 + (int)threadIdx.x;
    int row = idx >> 5;
    int colb = (idx & 31) * 8;
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if (mtail && grow >= M) continue;
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BN2 + colb + j];
    } else if (!ntail || gcol < N) {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BN2 + colb);
    }
  }
}



// ==== 256x256 / BK=32 three-buffer pipelined GEMM (v2) ==============
// The round-1 256-tile kernel drains vmcnt(0) at EVERY K-tile boundary
// (its __syncthreads carries the wait while LDS-DMA is in flight) and
// runs one barrier per phase; MfmaUtil measured 40%.  This kernel is
// built to the verified 8-phase-template structure instead:
//  * THREE LDS tile buffers (BK=32 so 3x(A+B) = 96 KiB fits): during
//    tile t the wave stages tile t+2, so at the t->t+1 boundary the
//    counted wait vmcnt(4) proves tile t+1 landed while t+2's four
//    DMAs stay in flight.  No vmcnt(0) anywhere in the main loop.
//  * TWO raw barriers per phase isolating the MFMA cluster (reads +
//    stage issue | barrier; lgkmcnt(0); setprio(1); 16 MFMA;
//    setprio(0) | barrier): on each SIMD one wave's MFMA segment
//    overlaps its partner's load segment.
//  * Conflict-free LDS k-slot permutation: LDS(row, slot') holds
//    k-chunk c at slot' = c ^ h(q), q = (row>>2)&3, h = [0,2,3,1].
//    ds_read_b128 services a wave in four fixed 16-lane groups that
//    MIX k-slots ({f0-3,f12-15}@s + {f4-11}@s+1); with this h every
//    group's 16 addresses hit the 16 bank-16-units exactly once
//    (checked against the gfx950 b128 lane-group table) — the plain
//    XOR swizzles leave 2-4-way conflicts on this pattern.  The
//    permutation commutes with the lane-linear glds write (source
//    column of lane l = ((l&3) ^ h((l>>4)&3))*16B) and each 4-lane row
//    group still covers one contiguous 64-B global segment.
constexpr int BK3 = 32;
constexpr int ROW3 = BK3 * 2;          // 64 B per LDS row
constexpr int TILE3 = 256 * ROW3;      // one operand tile = 16 KiB

__device__ __forceinline__ int hq3(int q) {       // [0,2,3,1]
  return (0x1320 >> (q * 4)) & 3;
}
__device__ __forceinline__ int rot3(int row, int slot) {
  return row * ROW3 + ((slot ^ hq3((row >> 2) & 3)) << 4);
}

// Stage one operand K-tile (256 rows x 32 bf16) into lds_img: 2 glds
// per wave (16 rows each x 8 waves = 256 rows per pass... 2 passes).
// Lane l of the stage covers (row r0 + (l>>2), slot l&3); the global
// column is the rotation-inverse so the LDS image equals rot3().
__device__ __forceinline__ void glds_stage32(
    const short* __restrict__ src, long ld, int row0, int rows_max,
    int k0, char* lds_img, int wid, int lane) {
  const int rl = lane >> 2;                      // row within 16-group
  const int gcolb = (((lane & 3) ^ hq3((lane >> 4) & 3)) << 4);
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    int rp = p * 128 + wid * 16;
    int gr = row0 + rp + rl;
    if (gr >= rows_max) gr = rows_max - 1;
    const void* g = src + (long)gr * ld + k0 + gcolb / 2;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)
            (lds_img + rp * ROW3),
        16, 0, 0);
  }
}

__global__ __launch_bounds__(NT2, 1) void gemm_nt256_v2_kernel(
    const short* __restrict__ A, const short* __restrict__ B,
    short* __restrict__ C, const short* __restrict__ bias,
    int M, int N, int K, int relu) {
  int nwg = gridDim.x;
  int wg = blockIdx.x;
  {
    int q = nwg / 8, rr = nwg % 8, xcd = wg % 8, idx = wg / 8;
    wg = (xcd < rr ? xcd * (q + 1) : rr * (q + 1) + (xcd - rr) * q) + idx;
  }
  const int nbn = (N + BN2 - 1) / BN2;
  const int bm0 = (wg / nbn) * BM2;
  const int bn0 = (wg % nbn) * BN2;

  // 3 x (A,B) staging buffers (96 KiB) inside the 128 KiB block that
  // the epilogue reuses as the C bounce image.  ONE __shared__ object
  // only (a second one makes hipcc emit vmcnt(0) before every ds_read
  // of a glds pipeline).
  __shared__ __attribute__((aligned(16))) char lds[2 * BM2 * BN2];
  // buffer b occupies [b*2*TILE3, b*2*TILE3 + 2*TILE3): A then B

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int wr = (wid >> 2) * 128;   // 2(M) x 4(N) wave grid
  const int wc = (wid & 3) * 64;

  float4v acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nk = K / BK3;
  // prologue: stage tiles 0 and 1; prove tile 0 landed (4 younger
  // DMAs stay in flight), then one full barrier.
  glds_stage32(A, K, bm0, M, 0, lds, wid, lane);
  glds_stage32(B, K, bn0, N, 0, lds + TILE3, wid, lane);
  if (nk > 1) {
    glds_stage32(A, K, bm0, M, BK3, lds + 2 * TILE3, wid, lane);
    glds_stage32(B, K, bn0, N, BK3, lds + 3 * TILE3, wid, lane);
  }
  // vmcnt(4): tile 0 landed, tile 1's four DMAs stay in flight
  if (nk > 1) __builtin_amdgcn_s_waitcnt(0xf70 | 4);
  else __builtin_amdgcn_s_waitcnt(0xf70);
  __builtin_amdgcn_s_barrier();

  const int frow = lane & 15;        // fragment row within 16
  const int slot = lane >> 4;        // k slot (16 B)

  for (int t = 0; t < nk; ++t) {
    const char* Act = lds + (t % 3) * (2 * TILE3);
    const char* Bct = Act + TILE3;
    char* nxt = lds + ((t + 2) % 3) * (2 * TILE3);
    const bool pre = t + 2 < nk;
    const int k2 = (t + 2) * BK3;
    bf16x8 bfr[4];
#pragma unroll
    for (int ph = 0; ph < 2; ++ph) {   // jr half of the wave's rows
      bf16x8 af[4];
#pragma unroll
      for (int fi = 0; fi < 4; ++fi) {
        int r = wr + ph * 64 + fi * 16 + frow;
        af[fi] = *reinterpret_cast<const bf16x8*>(Act + rot3(r, slot));
      }
      if (ph == 0) {
#pragma unroll
        for (int fj = 0; fj < 4; ++fj) {
          int r = wc + fj * 16 + frow;
          bfr[fj] = *reinterpret_cast<const bf16x8*>(Bct + rot3(r, slot));
        }
        if (pre)
          glds_stage32(A, K, bm0, M, k2, nxt, wid, lane);
      } else if (pre) {
        glds_stage32(B, K, bn0, N, k2, nxt + TILE3, wid, lane);
      }
      __builtin_amdgcn_s_barrier();
      asm volatile("s_waitcnt lgkmcnt(0)");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int fi = 0; fi < 4; ++fi)
#pragma unroll
        for (int fj = 0; fj < 4; ++fj)
          acc[ph * 4 + fi][fj] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af[fi], bfr[fj], acc[ph * 4 + fi][fj], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      if (ph == 1 && t + 1 < nk) {
        // tile boundary: prove tile t+1 landed; t+2's 4 DMAs (just
        // issued this tile) stay in flight across the barrier.
        if (pre) __builtin_amdgcn_s_waitcnt(0xf70 | 4);
        else __builtin_amdgcn_s_waitcnt(0xf70);
      }
      __builtin_amdgcn_s_barrier();
    }
  }

  const bool mtail = bm0 + BM2 > M;
  const bool ntail = bn0 + BN2 > N;
  // epilogue: LDS bounce (identical to gemm_nt256_kernel)
  __builtin_amdgcn_s_waitcnt(0xf70);   // drain any stray DMA
  __builtin_amdgcn_s_barrier();
  short* c_lds = (short*)lds;  // [BM2][BN2]
  {
    const int lrow0 = wr + (lane >> 4) * 4;
    const int lcol0 = wc + (lane & 15);
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        int col = lcol0 + j * 16;
        float bv = bias ? bf16_to_f32(bias[min(bn0 + col, N - 1)]) : 0.f;
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          float v = acc[i][j][r] + bv;
          if (relu) v = fmaxf(v, 0.f);
          c_lds[(lrow0 + i * 16 + r) * BN2 + col] = f32_to_bf16(v);
        }
      }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 16; ++p) {
    int idx = p * NT2 + (int)threadIdx.x;
    int row = idx >> 5;
    int colb = (idx & 31) * 8;
    int grow = bm0 + row;
    int gcol = bn0 + colb;
    if (mtail && grow >= M) continue;
    if (ntail && gcol + 7 >= N) {
      for (int j = 0; j < 8 && gcol + j < N; ++j)
        C[(long)grow * N + gcol + j] = c_lds[row * BN2 + colb + j];
    } else if (!ntail || gcol < N) {
      *reinterpret_cast<short8*>(C + (long)grow * N + gcol) =
          *reinterpret_cast<const short8*>(c_lds + row * BN2 + colb);
    }
  }
}

// ---- bf16 2-D transpose (LDS-tiled, vectorized loads) --------------
// grid.z = batch index (stride R*C per matrix) for batched use.
constexpr int TT = 64;  // tile
// LDS tile index with XOR bank swizzle: the former [TT][TT+8] pad was
// useless for this kernel's stride-8-row column reads (8 rows x 72
// shorts = 288 dwords = 0 mod 32 banks -> 8-way conflicts, 22% of
// GPUTime per rocprofv3 --pmc LDSBankConflict); xoring the column by
// (row&7)*8 keeps 16B store blocks intact and spreads the 8-row read
// column across 8 distinct banks.
__device__ __forceinline__ int tsw(int r, int c) {
  return r * TT + (c ^ ((r & 7) << 3));
}
// colsum (optional): fp32 per-input-column sums accumulated while the
// tile is already in LDS — gives nn.Linear's bias grad (dy.sum(0)) for
// free instead of a separate full-tensor torch reduce pass.
template <bool COLSUM>
__global__ __launch_bounds__(256) void transpose_kernel(
    const short* __restrict__ in0, short* __restrict__ out0,
    float* __restrict__ colsum, int R, int Cc) {
  const short* in = in0 + (long)blockIdx.z * R * Cc;
  short* out = out0 + (long)blockIdx.z * R * Cc;
  __shared__ short tile[TT * TT];
  int tr0 = blockIdx.y * TT, tc0 = blockIdx.x * TT;
  int tid = threadIdx.x;
  // load: 256 threads x 8 cols = 2048 elems/pass; 2 passes for 64x64
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    int r = p * 32 + (tid >> 3);
    int c = (tid & 7) * 8;
    int gr = tr0 + r, gc = tc0 + c;
    if (gr < R) {
      if (gc + 7 < Cc) {
        short8 v = *reinterpret_cast<const short8*>(in + (long)gr * Cc + gc);
        *reinterpret_cast<short8*>(tile + tsw(r, c)) = v;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          tile[tsw(r, c + j)] = (gc + j < Cc)
              ? in[(long)gr * Cc + gc + j] : 0;
      }
    }
  }
  __syncthreads();
  if (COLSUM) {
    // threads 0..63 each own one tile column; per-row-tile partials go
    // to slot row blockIdx.y (plain store — no atomics, and no
    // zero-fill of the accumulator before the launch); a tiny reduce
    // kernel folds the R/64 slot rows into the bf16 bias grad.
    if (tid < TT && tc0 + tid < Cc) {
      float s = 0.f;
      int rmax = min(TT, R - tr0);
      for (int r = 0; r < rmax; ++r) s += bf16_to_f32(tile[tsw(r, tid)]);
      colsum[(long)blockIdx.y * Cc + tc0 + tid] = s;
    }
  }
  // store: out[c][r]; thread writes 8 contiguous out-cols (= in-rows)
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    int c = p * 32 + (tid >> 3);   // in col = out row
    int r = (tid & 7) * 8;         // in row = out col
    int gor = tc0 + c, goc = tr0 + r;
    if (gor < Cc) {
      if (goc + 7 < R) {
        short8 v;
#pragma unroll
        for (int j = 0; j < 8; ++j) v[j] = tile[tsw(r + j, c)];
        *reinterpret_cast<short8*>(out + (long)gor * R + goc) = v;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (goc + j < R)
            out[(long)gor * R + goc + j] = tile[tsw(r + j, c)];
      }
    }
  }
}

// strided-input batched transpose: logical [B,H,R,C] tensor with
// arbitrary (b,h,row) strides (col stride 1) -> contiguous
// [B*H, C, R].  Lets attention pre-transpose Q/K/V/dO slices of a
// packed projection without materializing contiguous copies first.
__global__ __launch_bounds__(256) void transpose_bhsd_kernel(
    const short* __restrict__ in0, short* __restrict__ out0,
    int R, int Cc, int H, long sB, long sH, long sR) {
  const long z = blockIdx.z;
  const short* in = in0 + (z / H) * sB + (z % H) * sH;
  short* out = out0 + z * (long)R * Cc;
  __shared__ short tile[TT * TT];
  int tr0 = blockIdx.y * TT, tc0 = blockIdx.x * TT;
  int tid = threadIdx.x;
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    int r = p * 32 + (tid >> 3);
    int c = (tid & 7) * 8;
    int gr = tr0 + r, gc = tc0 + c;
    if (gr < R) {
      if (gc + 7 < Cc) {
        short8 v = *reinterpret_cast<const short8*>(
            in + (long)gr * sR + gc);
        *reinterpret_cast<short8*>(tile + tsw(r, c)) = v;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          tile[tsw(r, c + j)] = (gc + j < Cc)
              ? in[(long)gr * sR + gc + j] : 0;
      }
    }
  }
  __syncthreads();
#pragma unroll
  for (int p = 0; p < 2; ++p) {
    int c = p * 32 + (tid >> 3);
    int r = (tid & 7) * 8;
    int gor = tc0 + c, goc = tr0 + r;
    if (gor < Cc) {
      if (goc + 7 < R) {
        short8 v;
#pragma unroll
        for (int j = 0; j < 8; ++j) v[j] = tile[tsw(r + j, c)];
        *reinterpret_cast<short8*>(out + (long)gor * R + goc) = v;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j)
          if (goc + j < R)
            out[(long)gor * R + goc + j] = tile[tsw(r + j, c)];
      }
    }
  }
}

// slot partials [P][C] f32 -> bf16 column sums [C] (bias grad).
// 64 columns x 16 slot-lanes per 1024-thread block: the P loop is
// 16-way parallel per column (a serial P loop with C/256 blocks left
// 290+ CUs idle and ran 4x slower than the transpose it serves).
__global__ void colsum_reduce_kernel(const float* __restrict__ slots,
                                     short* __restrict__ db,
                                     int P, int C) {
  __shared__ float sh[64][16];
  const int cl = threadIdx.x & 63, pl = threadIdx.x >> 6;
  const int c = blockIdx.x * 64 + cl;
  float s = 0.f;
  if (c < C)
    for (int p = pl; p < P; p += 16) s += slots[(long)p * C + c];
  sh[cl][pl] = s;
  __syncthreads();
  if (pl != 0 || c >= C) return;
  s = 0.f;
#pragma unroll
  for (int j = 0; j < 16; ++j) s += sh[cl][j];
  db[c] = f32_to_bf16(s);
}

}  // namespace

torch::Tensor gemm_nt(torch::Tensor a, torch::Tensor b,
                      c10::optional<torch::Tensor> bias, bool relu,
                      c10::optional<torch::Tensor> add) {
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && b.is_contiguous());
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
              b.scalar_type() == torch::kBFloat16, "gemm_nt: bf16 only");
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2, "gemm_nt: 2-D only");
  const int M = a.size(0), K = a.size(1), N = b.size(0);
  TORCH_CHECK(b.size(1) == K, "gemm_nt: K mismatch");
  TORCH_CHECK(K % BK == 0, "gemm_nt: K must be a multiple of ", BK);
  auto c = torch::empty({M, N}, a.options());
  const short* bias_p = nullptr;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->scalar_type() == torch::kBFloat16 &&
                bias->numel() == N);
    bias_p = (const short*)bias->data_ptr();
  }
  // optional fused elementwise addend (residual stream): supported by
  // the p8 / p8-128 default kernels; gated experimental variants fall
  // back to a separate add below.
  const short* add_p = nullptr;
  if (add.has_value() && add->defined()) {
    TORCH_CHECK(add->scalar_type() == torch::kBFloat16 &&
                add->is_contiguous() && add->sizes() == c.sizes());
    TORCH_CHECK(!relu, "gemm_nt: relu and fused addend are exclusive "
                       "(the fused path applies add BEFORE relu, the "
                       "fallback after — no caller composes them)");
    add_p = (const short*)add->data_ptr();
  }
  auto finish = [&](torch::Tensor out) {  // fallback for variants
    if (add_p) out.add_(*add);             // without the fused addend
    return out;
  };
  auto stream = c10::cuda::getCurrentCUDAStream();
  // 256-tile kernel runs 1 block/CU (128 KiB LDS): only profitable
  // when its grid fills most of the 256 CUs; otherwise the 128-tile
  // kernel's 4x block count wins.
  long blocks256 = (long)((M + BM2 - 1) / BM2) * ((N + BN2 - 1) / BN2);
  static const bool dbg = [] {
    const char* e = getenv("GSAMD_GEMM_DEBUG");
    return e && e[0] == '1';
  }();
  if (dbg)
    fprintf(stderr, "gemm_nt M=%d N=%d K=%d blocks256=%ld path=%s\n",
            M, N, K, blocks256,
            (M >= 256 && N >= 256 && blocks256 >= 200) ? "256" : "128");
  if (M >= 256 && N >= 256 && blocks256 >= 200) {
    int grid = (int)blocks256;
    // variant select: 0 = round-1 kernel (vmcnt(0) drain per tile),
    // 1 = v2 (BK=32, 3 buffers), 2 = p8 (BK=64, consumption-staged
    // counted-vmcnt pipeline) — the measured default.
    static const int var = [] {
      const char* e = getenv("GSAMD_GEMM_VAR");
      return e ? atoi(e) : 7;   // p8 + static wave priority: the
                                // measured winner (profiles/README.md)
    }();
    if (var == 6) {
      hipLaunchKernelGGL(gemm_nt256_p12_kernel, dim3(grid), dim3(NT2), 0,
                         stream.stream(), (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, M, N, K, relu ? 1 : 0);
      HIP_CHECK_LAST();
      return finish(c);
    }
    if (var == 5) {
      hipLaunchKernelGGL(gemm_nt256_p11_kernel, dim3(grid), dim3(NT2), 0,
                         stream.stream(), (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, M, N, K, relu ? 1 : 0);
      HIP_CHECK_LAST();
      return finish(c);
    }
    if (var == 4) {
      hipLaunchKernelGGL(gemm_nt256_p10_kernel, dim3(grid), dim3(NT2), 0,
                         stream.stream(), (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, M, N, K, relu ? 1 : 0);
      HIP_CHECK_LAST();
      return finish(c);
    }
    if (var == 3) {
      hipLaunchKernelGGL(gemm_nt256_p9_kernel, dim3(grid), dim3(NT2), 0,
                         stream.stream(), (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, M, N, K, relu ? 1 : 0);
      HIP_CHECK_LAST();
      return finish(c);
    }
    if (var == 7) {
      hipLaunchKernelGGL((gemm_nt256_p8_kernel<true>), dim3(grid),
                         dim3(NT2), 0,
                         stream.stream(), (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, add_p, M, N, K, relu ? 1 : 0);
      HIP_CHECK_LAST();
      return c;
    }
    if (var == 2) {
      hipLaunchKernelGGL((gemm_nt256_p8_kernel<false>), dim3(grid),
                         dim3(NT2), 0,
                         stream.stream(), (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, add_p, M, N, K, relu ? 1 : 0);
      HIP_CHECK_LAST();
      return c;
    }
    if (var == 1 && K % BK3 == 0) {
      hipLaunchKernelGGL(gemm_nt256_v2_kernel, dim3(grid), dim3(NT2), 0,
                         stream.stream(), (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, M, N, K, relu ? 1 : 0);
      HIP_CHECK_LAST();
      return finish(c);
    }
    // consumption-scheduled staging variant (counted vmcnt waits, no
    // boundary drain) — A/B gate while it is being qualified
    static const bool use_cs = [] {
      const char* e = getenv("GSAMD_GEMM_CS");
      return e && e[0] == '1';
    }();
    if (use_cs) {
      hipLaunchKernelGGL(gemm_nt256_cs_kernel, dim3(grid), dim3(NT2), 0,
                         stream.stream(), (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, M, N, K, relu ? 1 : 0);
    } else {
      hipLaunchKernelGGL(gemm_nt256_kernel, dim3(grid), dim3(NT2), 0,
                         stream.stream(), (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, M, N, K, relu ? 1 : 0);
    }
  } else {
    // half-width 256x128 p8 for mid grids (llama batch-1: M=2048)
    // MEASURED: ties the 128-tile path at big-K M=2048 shapes (+0-3%)
    // and LOSES ~15% at small-K (BERT, K=768: 1 block/CU cannot beat
    // the 128-tile's 2 blocks/CU when the prologue dominates) -> off
    // by default, kept as a gated experiment.
    static const bool use_h = [] {
      const char* e = getenv("GSAMD_GEMM_H");
      return e && e[0] == '1';
    }();
    long blocksH = (long)((M + BM2 - 1) / BM2) * ((N + 127) / 128);
    if (use_h && M >= 256 && N >= 128 && blocksH >= 200) {
      hipLaunchKernelGGL(gemm_nt256h_p8_kernel, dim3((int)blocksH),
                         dim3(NT2), 0, stream.stream(),
                         (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, M, N, K, relu ? 1 : 0);
      HIP_CHECK_LAST();
      return finish(c);
    }
    // small-grid default: the p8-discipline 128-tile kernel (2
    // blocks/CU); GSAMD_GEMM128=0 falls back to the round-1 kernel.
    static const int use128p8 = [] {
      const char* e = getenv("GSAMD_GEMM128");
      return !e || e[0] != '0';
    }();
    static const bool use32 = [] {
      const char* e = getenv("GSAMD_GEMM32");
      return e && e[0] == '1';
    }();
    int grid = ((M + BM - 1) / BM) * ((N + BN - 1) / BN);
    if (use128p8 && !use32) {
      hipLaunchKernelGGL(gemm_nt128_p8_kernel, dim3(grid), dim3(NTHREADS),
                         0, stream.stream(), (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, add_p, M, N, K, relu ? 1 : 0);
      HIP_CHECK_LAST();
      return c;
    }
    if (use32) {
      hipLaunchKernelGGL(gemm_nt32_kernel, dim3(grid), dim3(NTHREADS), 0,
                         stream.stream(), (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, M, N, K, relu ? 1 : 0);
    } else {
      hipLaunchKernelGGL(gemm_nt_kernel, dim3(grid), dim3(NTHREADS), 0,
                         stream.stream(), (const short*)a.data_ptr(),
                         (const short*)b.data_ptr(), (short*)c.data_ptr(),
                         bias_p, M, N, K, relu ? 1 : 0);
    }
  }
  HIP_CHECK_LAST();
  return finish(c);
}

// C[M,N] = A[M,R] . B[R,N] with B stored red-major (dX = dY . W with
// W exactly as stored — no transpose pass).  256-tile only: the caller
// gates small grids to the transpose+NT path.  R must be %64 (it is
// the reduction dim; clamping would corrupt the product).
torch::Tensor gemm_nn(torch::Tensor a, torch::Tensor b,
                      c10::optional<torch::Tensor> bias, bool relu) {
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && b.is_contiguous());
  TORCH_CHECK(a.scalar_type() == torch::kBFloat16 &&
              b.scalar_type() == torch::kBFloat16, "gemm_nn: bf16 only");
  TORCH_CHECK(a.dim() == 2 && b.dim() == 2, "gemm_nn: 2-D only");
  const int M = a.size(0), R = a.size(1), N = b.size(1);
  TORCH_CHECK(b.size(0) == R, "gemm_nn: reduction dim mismatch");
  TORCH_CHECK(R % BK2 == 0, "gemm_nn: R must be a multiple of ", BK2);
  TORCH_CHECK(N % 8 == 0, "gemm_nn: N must be a multiple of 8");
  auto c = torch::empty({M, N}, a.options());
  const short* bias_p = nullptr;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->scalar_type() == torch::kBFloat16 &&
                bias->numel() == N);
    bias_p = (const short*)bias->data_ptr();
  }
  long grid = (long)((M + BM2 - 1) / BM2) * ((N + BN2 - 1) / BN2);
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(gemm_nn_kernel, dim3((int)grid), dim3(NT2), 0,
                     stream.stream(), (const short*)a.data_ptr(),
                     (const short*)b.data_ptr(), (short*)c.data_ptr(),
                     bias_p, M, N, R, relu ? 1 : 0);
  HIP_CHECK_LAST();
  return c;
}

torch::Tensor transpose2d(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(x.dim() == 2 || x.dim() == 3,
              "transpose2d: 2-D or batched 3-D");
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int batched = x.dim() == 3;
  const long Bn = batched ? x.size(0) : 1;
  const int R = x.size(batched ? 1 : 0), C = x.size(batched ? 2 : 1);
  auto out = batched ? torch::empty({Bn, C, R}, x.options())
                     : torch::empty({C, R}, x.options());
  dim3 grid((C + TT - 1) / TT, (R + TT - 1) / TT, Bn);
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(transpose_kernel<false>, grid, dim3(256), 0,
                     stream.stream(), (const short*)x.data_ptr(),
                     (short*)out.data_ptr(), nullptr, R, C);
  HIP_CHECK_LAST();
  return out;
}

// strided 4-D transpose: x logical [B,H,S,D] (any strides, D-dim
// contiguous) -> contiguous [B,H,D,S]
torch::Tensor transpose2d_bhsd(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 4 && x.stride(3) == 1);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int B = x.size(0), H = x.size(1), S = x.size(2), D = x.size(3);
  auto out = torch::empty({B, H, D, S}, x.options());
  dim3 grid((D + TT - 1) / TT, (S + TT - 1) / TT, B * H);
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(transpose_bhsd_kernel, grid, dim3(256), 0,
                     stream.stream(), (const short*)x.data_ptr(),
                     (short*)out.data_ptr(), S, D, H,
                     x.stride(0), x.stride(1), x.stride(2));
  HIP_CHECK_LAST();
  return out;
}

// transpose + column sums (dy -> dy^T and bf16 bias grad in one pass)
std::vector<torch::Tensor> transpose2d_colsum(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 2);
  TORCH_CHECK(x.scalar_type() == torch::kBFloat16);
  const int R = x.size(0), C = x.size(1);
  auto out = torch::empty({C, R}, x.options());
  dim3 grid((C + TT - 1) / TT, (R + TT - 1) / TT, 1);
  auto slots = torch::empty({(int)grid.y, C},
                            x.options().dtype(torch::kFloat32));
  auto db = torch::empty({C}, x.options());
  auto stream = c10::cuda::getCurrentCUDAStream();
  hipLaunchKernelGGL(transpose_kernel<true>, grid, dim3(256), 0,
                     stream.stream(), (const short*)x.data_ptr(),
                     (short*)out.data_ptr(), slots.data_ptr<float>(),
                     R, C);
  hipLaunchKernelGGL(colsum_reduce_kernel, dim3((C + 63) / 64),
                     dim3(1024), 0, stream.stream(),
                     slots.data_ptr<float>(), (short*)db.data_ptr(),
                     (int)grid.y, C);
  HIP_CHECK_LAST();
  return {out, db};
}
