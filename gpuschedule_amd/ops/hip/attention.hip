// Flash-attention forward for gfx950 (MI355X): online-softmax MFMA.
//
// Layout strategy (all fragments reuse the 16x16x32 bf16 MFMA mappings
// verified by the GEMM kernel):
//  * Q stays in registers: each wave owns 16 query rows; A-fragments
//    read straight from global (row = lane&15, 8 bf16 along D).
//  * K tile [KVB=64][D] staged into LDS by direct HBM->LDS DMA
//    (global_load_lds, source-swizzled columns) with the GEMM XOR
//    swizzle; QK^T B-fragments are the same addressing as the GEMM B
//    operand.  Double-buffered: the next tile's DMA issues before the
//    current tile's compute (async-split), drained at the tile
//    barrier.
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
// One workgroup = 4 (or 8, long-seq gate) independent waves = 64/128
// query rows of one (b, h).
// Causal masking: whole KV tiles after the diagonal are skipped;
// diagonal tiles mask per element.  Sequence tails are masked to -inf.
// Q/K and O go through (batch, head, row) stride triples so packed-QKV
// projection slices run with zero layout copies; the contiguous case
// constant-folds the strides via the STRIDED template split.
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

// Q/K and O are addressed through (batch, head, row) stride triples so
// strided views work directly — e.g. the packed-QKV path (q/k/v are
// slices of one [B,S,3,H,D] projection, O written as a [B,S,H,D] view)
// runs with ZERO layout copies.  Contiguous [B,H,S,D] is the special
// case (sB=H*S*D, sH=S*D, sS=D).  Vt is an internal contiguous
// scratch either way.
template <int D, bool CAUSAL, int NW, bool STRIDED>
__global__ __launch_bounds__(NW * 64, 2) void attn_fwd_kernel(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ Vt, short* __restrict__ O,
    float* __restrict__ LSE, int B, int H, int HK, int S, float scale,
    long qsB, long qsH, long qsS_, long osB, long osH, long osS_) {
  // HK = kv heads (GQA: q head h reads kv head h/(H/HK) — stride-0
  // style sharing, no host-side K/V expansion).  The STRIDED
  // (packed-QKV) path requires HK == H (host-checked).
  // !STRIDED folds the row strides to the compile-time D (measured:
  // runtime-stride address math cost 8% on the 8-wave long-seq path)
  const long qsS = STRIDED ? qsS_ : (long)D;
  const long osS = STRIDED ? osS_ : (long)D;
  constexpr int NT = NW * 64;      // threads (NW waves x 16 q rows)
  constexpr int QB = NW * 16;      // q rows per block
  constexpr int DF = D / 16;       // PV col fragments
  constexpr int KC = D / 32;       // QK^T k-steps over D
  constexpr int ROWB = D * 2;      // K tile LDS row bytes
  // LDS: double-buffered K [KVB][D] + Vt [D][KVB] + NW wave-private P
  __shared__ __attribute__((aligned(16))) char lds[
      2 * (KVB * D * 2 + D * KVB * 2) + NW * 16 * KVB * 2];
  char* k_cur = lds;
  char* v_cur = lds + KVB * D * 2;
  char* k_nxt = lds + 2 * KVB * D * 2;
  char* v_nxt = lds + 3 * KVB * D * 2;
  char* p_lds = lds + 4 * KVB * D * 2;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int tid = threadIdx.x;
  char* my_p = p_lds + wid * 16 * KVB * 2;

  const int n_qt_g = (S + QB - 1) / QB;
  const int bh = blockIdx.x / n_qt_g;
  // causal: later q-tiles do more kv work; launch them FIRST so the
  // long blocks don't form the tail of the wave
  int qt = blockIdx.x % n_qt_g;
  if (CAUSAL) qt = n_qt_g - 1 - qt;
  const int b = bh / H, h = bh % H;
  const int q0 = qt * QB + wid * 16;      // this wave's first q row

  const long qb = (long)b * qsB + (long)h * qsH;
  const long ob = (long)b * osB + (long)h * osH;
  const int kh = h / (H / HK);                    // GQA kv head
  const long kb = STRIDED ? qb : ((long)(b * HK + kh) * S) * (long)D;
  const long voff = ((long)(b * HK + kh) * D) * S;  // Vt is [D][S]

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
            Q + qb + (long)row * qsS + d0);
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

  const int q_tile_end = qt * QB + QB - 1;
  const int n_kv = CAUSAL
      ? min((q_tile_end / KVB) + 1, (S + KVB - 1) / KVB)
      : (S + KVB - 1) / KVB;

  // ---- staging: direct HBM->LDS DMA (global_load_lds, 16 B/lane) --
  // Replaces the earlier register-staged T14 path: same async timing
  // (issue early, completion drained by the end-of-tile barrier) but
  // no kst/vst registers and no ds_write pass.  The XOR bank swizzle
  // moves to the SOURCE column (lane-linear LDS dest), exactly as in
  // the GEMM staging.  Tail rows/cols are CLAMPED to the last valid
  // position instead of zeroed: clamped K rows produce finite garbage
  // scores on columns the softmax kills with -inf, and clamped V
  // columns are multiplied by those zero probabilities.
  constexpr int PIECES_K = KVB * D * 2 / (NT * 16);
  constexpr int CPR = D * 2 / 16;     // 16B chunks per K row
  constexpr int PIECES_V = D * KVB * 2 / (NT * 16);
  constexpr int CPRV = KVB * 2 / 16;

  auto stage_dma = [&](int kv0, char* kbuf, char* vbuf) {
    const long kv_add = (long)kv0 * qsS;
#pragma unroll
    for (int p = 0; p < PIECES_K; ++p) {
      int idx = p * NT + tid;
      int r = idx / CPR;
      int cbs = ((idx % CPR) * 16) ^ ((r & 7) << 4);  // src-swizzled
      int kvr = kv0 + r;
      long roff = (kvr < S) ? kv_add + (long)r * qsS
                            : (long)(S - 1) * qsS;
      const void* g = K + kb + roff + cbs / 2;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)g,
          (__attribute__((address_space(3))) unsigned int*)
              (kbuf + (p * NT + (tid & ~63)) * 16),
          16, 0, 0);
    }
#pragma unroll
    for (int p = 0; p < PIECES_V; ++p) {
      int idx = p * NT + tid;
      int r = idx / CPRV;
      int cbs = ((idx % CPRV) * 16) ^ ((r & 7) << 4);
      int kvc = kv0 + cbs / 2;
      if (kvc + 7 >= S) kvc = S - 8;
      const void* g = Vt + voff + (long)r * S + kvc;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)g,
          (__attribute__((address_space(3))) unsigned int*)
              (vbuf + (p * NT + (tid & ~63)) * 16),
          16, 0, 0);
    }
  };

  stage_dma(0, k_cur, v_cur);
  __syncthreads();

  for (int t = 0; t < n_kv; ++t) {
    const int kv0 = t * KVB;
    char* k_lds = k_cur;
    char* v_lds = v_cur;
    // issue next tile's HBM->LDS DMA NOW; its latency hides under
    // QK^T + softmax + PV; the end-of-tile __syncthreads drains it
    if (t + 1 < n_kv) stage_dma((t + 1) * KVB, k_nxt, v_nxt);

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
    if (t + 1 < n_kv) {
      char* tmp = k_cur; k_cur = k_nxt; k_nxt = tmp;
      tmp = v_cur; v_cur = v_nxt; v_nxt = tmp;
    }
    __syncthreads();
  }

  // ---- epilogue: O / l, plus per-row logsumexp for backward -----
  const int orow0 = q0 + (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = orow0 + r;
    if (row >= S) continue;
    float inv = (l[r] > 0.f) ? 1.0f / l[r] : 0.f;
    short* orow = O + ob + (long)row * osS;
#pragma unroll
    for (int f = 0; f < DF; ++f)
      orow[f * 16 + (lane & 15)] = f32_to_bf16(o_acc[f][r] * inv);
    if (LSE && (lane & 15) == 0)
      LSE[(long)(b * H + h) * S + row] =
          (l[r] > 0.f) ? m[r] + __logf(l[r]) : -1e30f;
  }
}

// ==================== backward =====================================
// FA2-style two-kernel backward (no atomics):
//   bwd_kv: grid over KV tiles, accumulates dK/dV looping q-tiles;
//   bwd_q:  grid over Q tiles, accumulates dQ looping kv-tiles.
// All matmuls are the SAME validated NT MFMA pattern; operands whose
// reduction dim is the row dim (Q/K/dO in the dK/dV/dQ products) come
// from batched pre-transposed copies (Qt/Kt/dOt, one fast pass each),
// so every B-fragment reads 8 bf16 contiguous along the reduction.
// P / dS convert C-layout -> A-fragment layout through wave-private
// swizzled LDS, as in forward.  Definitions (scale folded into the
// staged K/Q fragments so exp uses the saved scaled-score lse):
//   P  = exp(S - lse[q]);  dP = dO . V^T;  Drow = rowsum(dO*O)
//   dS = P * (dP - Drow[q]) * scale
//   dV = P^T dO;  dK = dS^T Q;  dQ = dS K

// generic glds tile stage: ROWS x ROWBYTES image, 4 waves, source-
// swizzled so the linear LDS image is the swizzled image.  Row AND
// column clamped (clamped lanes load valid-but-wrong data; consumers
// mask those rows/cols to zero before they reach any stored output).
template <int ROWS, int ROWBYTES>
__device__ __forceinline__ void glds_tile(
    const short* __restrict__ src, long ld, int row0, int rows_max,
    int col0, int cols_max, char* img, int wid, int lane) {
  constexpr int RPI = 1024 / ROWBYTES;     // rows per instruction
  constexpr int CPR = ROWBYTES / 16;       // 16B chunks per row
  constexpr int NI = (ROWS / 4) / RPI;     // instructions per wave
  const int rl = lane / CPR;
  const int cb = (lane % CPR) * 16;
  // hoist the 64-bit row*ld multiplies out of the issue loop (ld is a
  // runtime stride now): one mul up front, adds of RPI*ld per piece,
  // and a precomputed uniform clamp offset for the rare tail rows
  const int rp0 = wid * (ROWS / 4) + rl;
  const long addr0 = (long)(row0 + rp0) * ld;
  const long step = (long)RPI * ld;
  const long clampoff = (long)(rows_max - 1) * ld;
#pragma unroll
  for (int p = 0; p < NI; ++p) {
    int rp = rp0 + p * RPI;
    int gr = row0 + rp;
    long a = (gr >= rows_max) ? clampoff : addr0 + p * step;
    int scol = (cb ^ ((rp & 7) << 4)) / 2 + col0;
    if (scol > cols_max - 8) scol = cols_max - 8;
    const void* g = src + a + scol;
    __builtin_amdgcn_global_load_lds(
        (const __attribute__((address_space(1))) unsigned int*)g,
        (__attribute__((address_space(3))) unsigned int*)
            (img + (wid * (ROWS / 4) + p * RPI) * ROWBYTES),
        16, 0, 0);
  }
}

// Drow = rowsum(dO * O): one 64-lane wave per row.  Drow stays in
// (b, h, s) order regardless of the O/dO layout (stride triple).
__global__ __launch_bounds__(256) void attn_drow_kernel(
    const short* __restrict__ dO, const short* __restrict__ O,
    float* __restrict__ Drow, long n_rows, int H, int S, int D,
    long osB, long osH, long osS) {
  long row = (long)blockIdx.x * 4 + (threadIdx.x >> 6);
  if (row >= n_rows) return;
  const int lane = threadIdx.x & 63;
  const int s_i = (int)(row % S);
  const long bh = row / S;
  const long base = (bh / H) * osB + (bh % H) * osH + (long)s_i * osS;
  float s = 0.f;
  for (int i = lane * 2; i + 1 < D; i += 128) {
    s += bf16_to_f32(dO[base + i]) * bf16_to_f32(O[base + i]);
    s += bf16_to_f32(dO[base + i + 1]) * bf16_to_f32(O[base + i + 1]);
  }
  s = wave_reduce_sum(s);
  if (lane == 0) Drow[row] = s;
}

template <int D, bool CAUSAL, bool STRIDED>
__global__ __launch_bounds__(NTHREADS, 2) void attn_bwd_kv_kernel(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, const short* __restrict__ dO,
    const short* __restrict__ Qt, const short* __restrict__ dOt,
    const float* __restrict__ LSE, const float* __restrict__ DR,
    short* __restrict__ dK, short* __restrict__ dV,
    int B, int H, int HK, int S, float scale,
    long qsB, long qsH, long qsS_, long osB, long osH, long osS_) {
  const long qsS = STRIDED ? qsS_ : (long)D;
  const long osS = STRIDED ? osS_ : (long)D;
  constexpr int DF = D / 16;
  constexpr int KC = D / 32;
  constexpr int RB = D * 2;
  __shared__ __attribute__((aligned(16))) char lds[
      2 * KVB * D * 2 +      // Q, dO  [64][D]
      2 * D * KVB * 2 +      // Qt, dOt [D][64]
      4 * 2 * 16 * KVB * 2 + // per-wave P^T and dS^T  [16][64]
      2 * KVB * 4];          // lse, Drow (f32[64])
  char* q_lds = lds;
  char* do_lds = q_lds + KVB * D * 2;
  char* qt_lds = do_lds + KVB * D * 2;
  char* dot_lds = qt_lds + D * KVB * 2;
  char* p_base = dot_lds + D * KVB * 2;
  float* lse_lds = (float*)(p_base + 4 * 2 * 16 * KVB * 2);
  float* dr_lds = lse_lds + KVB;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int tid = threadIdx.x;
  char* my_p = p_base + wid * 16 * KVB * 2;
  char* my_ds = p_base + (4 + wid) * 16 * KVB * 2;

  // grid is over (b, KV head, kv tile); for GQA the block loops the
  // rep = H/HK query heads that share this KV head, accumulating
  // dK/dV across the whole group IN REGISTERS — grouped (not atomic)
  // accumulation: deterministic, K/V loaded once per group.
  const int n_kvt = (S + KVB - 1) / KVB;
  const int bh = blockIdx.x / n_kvt;
  const int kvt = blockIdx.x % n_kvt;
  const int b = bh / HK, kvh = bh % HK;
  const int rep = H / HK;
  const int kv0 = kvt * KVB;
  const int krow = kv0 + wid * 16 + (lane & 15);

  const long kvb = STRIDED ? ((long)b * qsB + (long)kvh * qsH)
      : ((long)(b * HK + kvh) * S) * (long)D;   // K/V + dK/dV base

  // K (scaled) and V fragments in registers
  bf16x8 kreg[KC], vreg[KC];
  {
    int row = krow < S ? krow : S - 1;
#pragma unroll
    for (int c = 0; c < KC; ++c) {
      int d0 = c * 32 + (lane >> 4) * 8;
      bf16x8 kv8 = *reinterpret_cast<const bf16x8*>(
          K + kvb + (long)row * qsS + d0);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        kv8[j] = (__bf16)((float)kv8[j] * scale);
      kreg[c] = kv8;
      vreg[c] = *reinterpret_cast<const bf16x8*>(
          V + kvb + (long)row * qsS + d0);
    }
  }

  float4v dv_acc[DF], dk_acc[DF];
#pragma unroll
  for (int f = 0; f < DF; ++f) {
    dv_acc[f] = {0.f, 0.f, 0.f, 0.f};
    dk_acc[f] = {0.f, 0.f, 0.f, 0.f};
  }

  const int qt_start = CAUSAL ? kv0 / 64 : 0;
  const int n_qt = (S + 63) / 64;
  for (int g = 0; g < rep; ++g) {
  const int h = kvh * rep + g;
  const long qvb = (long)b * qsB + (long)h * qsH;   // Q base
  const long dob = (long)b * osB + (long)h * osH;   // dO base
  const long offT = ((long)(b * H + h) * D) * S;
  const long roff = (long)(b * H + h) * S;
  for (int qt = qt_start; qt < n_qt; ++qt) {
    const int q0 = qt * 64;
    glds_tile<KVB, RB>(Q + qvb, qsS, q0, S, 0, D, q_lds, wid, lane);
    glds_tile<KVB, RB>(dO + dob, osS, q0, S, 0, D, do_lds, wid, lane);
    glds_tile<D, KVB * 2>(Qt + offT, S, 0, D, q0, S, qt_lds, wid, lane);
    glds_tile<D, KVB * 2>(dOt + offT, S, 0, D, q0, S, dot_lds, wid, lane);
    for (int i = tid; i < KVB; i += NTHREADS) {
      int qr = q0 + i;
      lse_lds[i] = qr < S ? LSE[roff + qr] : 3e30f;
      dr_lds[i] = qr < S ? DR[roff + qr] : 0.f;
    }
    __syncthreads();

    // S^T[16k][64q] and dP^T[16k][64q]
    float4v st[4], dpt[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      st[f] = {0.f, 0.f, 0.f, 0.f};
      dpt[f] = {0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int kk = 0; kk < KC; ++kk) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 qf = *reinterpret_cast<const bf16x8*>(
            q_lds + swz(f * 16 + (lane & 15),
                        kk * 64 + (lane >> 4) * 16, RB));
        st[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            kreg[kk], qf, st[f], 0, 0, 0);
        bf16x8 dof = *reinterpret_cast<const bf16x8*>(
            do_lds + swz(f * 16 + (lane & 15),
                         kk * 64 + (lane >> 4) * 16, RB));
        dpt[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            vreg[kk], dof, dpt[f], 0, 0, 0);
      }
    }
    // P^T, dS^T -> wave-private LDS
    const int krow0 = kv0 + wid * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int qcol = q0 + f * 16 + (lane & 15);
      float lse_q = lse_lds[f * 16 + (lane & 15)];
      float dr_q = dr_lds[f * 16 + (lane & 15)];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        bool dead = qcol >= S || lse_q > 1e30f;
        if (CAUSAL) dead |= qcol < krow0 + r;
        float p = dead ? 0.f : __expf(st[f][r] - lse_q);
        float ds = p * (dpt[f][r] - dr_q) * scale;
        int row = (lane >> 4) * 4 + r;
        int cb = (f * 16 + (lane & 15)) * 2;
        *reinterpret_cast<short*>(
            my_p + swz(row, cb, KVB * 2)) = f32_to_bf16(p);
        *reinterpret_cast<short*>(
            my_ds + swz(row, cb, KVB * 2)) = f32_to_bf16(ds);
      }
    }
    // dV += P^T dO (B from dOt image), dK += dS^T Q (B from Qt image)
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 pf = *reinterpret_cast<const bf16x8*>(
          my_p + swz(lane & 15, kk * 64 + (lane >> 4) * 16, KVB * 2));
      bf16x8 dsf = *reinterpret_cast<const bf16x8*>(
          my_ds + swz(lane & 15, kk * 64 + (lane >> 4) * 16, KVB * 2));
#pragma unroll
      for (int f = 0; f < DF; ++f) {
        bf16x8 dotf = *reinterpret_cast<const bf16x8*>(
            dot_lds + swz(f * 16 + (lane & 15),
                          kk * 64 + (lane >> 4) * 16, KVB * 2));
        dv_acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pf, dotf, dv_acc[f], 0, 0, 0);
        bf16x8 qtf = *reinterpret_cast<const bf16x8*>(
            qt_lds + swz(f * 16 + (lane & 15),
                         kk * 64 + (lane >> 4) * 16, KVB * 2));
        dk_acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsf, qtf, dk_acc[f], 0, 0, 0);
      }
    }
    __syncthreads();
  }
  }  // q-head group

  const int orow0 = kv0 + wid * 16 + (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = orow0 + r;
    if (row >= S) continue;
    short* dkr = dK + kvb + (long)row * qsS;
    short* dvr = dV + kvb + (long)row * qsS;
#pragma unroll
    for (int f = 0; f < DF; ++f) {
      dkr[f * 16 + (lane & 15)] = f32_to_bf16(dk_acc[f][r]);
      dvr[f * 16 + (lane & 15)] = f32_to_bf16(dv_acc[f][r]);
    }
  }
}

// Pipelined dK/dV backward for D=128.  The base kernel's LDS image
// (Q,dO,Qt,dOt + per-wave P AND dS) is 96.5 KiB -> ONE block/CU, ONE
// wave/SIMD: nothing hides the per-iteration cold vmcnt(0) drain of
// tiles staged immediately before the barrier.  This variant:
//  * aliases dS onto the P buffer (write P -> consume P for dV ->
//    write dS over it -> consume for dK; wave-private, lgkm-ordered),
//  * double-buffers the staged set with the freed 16 KiB (145 KiB
//    total, still 1 block/CU),
//  * flattens the (q-head group x q-tile) loops and issues each
//    iteration's DMA a FULL iteration early — the one __syncthreads
//    per iteration then drains loads that are ~an iteration old.
template <int D, bool CAUSAL, bool STRIDED>
__global__ __launch_bounds__(NTHREADS, 2) void attn_bwd_kv_pipe_kernel(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, const short* __restrict__ dO,
    const short* __restrict__ Qt, const short* __restrict__ dOt,
    const float* __restrict__ LSE, const float* __restrict__ DR,
    short* __restrict__ dK, short* __restrict__ dV,
    int B, int H, int HK, int S, float scale,
    long qsB, long qsH, long qsS_, long osB, long osH, long osS_) {
  const long qsS = STRIDED ? qsS_ : (long)D;
  const long osS = STRIDED ? osS_ : (long)D;
  constexpr int DF = D / 16;
  constexpr int KC = D / 32;
  constexpr int RB = D * 2;
  constexpr int STG = 2 * KVB * D * 2 + 2 * D * KVB * 2 + 2 * KVB * 4;
  __shared__ __attribute__((aligned(16))) char lds[
      2 * STG + 4 * 16 * KVB * 2];
  char* p_base = lds + 2 * STG;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int tid = threadIdx.x;
  char* my_p = p_base + wid * 16 * KVB * 2;  // P, then reused for dS

  const int n_kvt = (S + KVB - 1) / KVB;
  const int bh = blockIdx.x / n_kvt;
  const int kvt = blockIdx.x % n_kvt;
  const int b = bh / HK, kvh = bh % HK;
  const int rep = H / HK;
  const int kv0 = kvt * KVB;
  const int krow = kv0 + wid * 16 + (lane & 15);

  const long kvb = STRIDED ? ((long)b * qsB + (long)kvh * qsH)
      : ((long)(b * HK + kvh) * S) * (long)D;

  bf16x8 kreg[KC], vreg[KC];
  {
    int row = krow < S ? krow : S - 1;
#pragma unroll
    for (int c = 0; c < KC; ++c) {
      int d0 = c * 32 + (lane >> 4) * 8;
      bf16x8 kv8 = *reinterpret_cast<const bf16x8*>(
          K + kvb + (long)row * qsS + d0);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        kv8[j] = (__bf16)((float)kv8[j] * scale);
      kreg[c] = kv8;
      vreg[c] = *reinterpret_cast<const bf16x8*>(
          V + kvb + (long)row * qsS + d0);
    }
  }

  float4v dv_acc[DF], dk_acc[DF];
#pragma unroll
  for (int f = 0; f < DF; ++f) {
    dv_acc[f] = {0.f, 0.f, 0.f, 0.f};
    dk_acc[f] = {0.f, 0.f, 0.f, 0.f};
  }

  const int qt_start = CAUSAL ? kv0 / 64 : 0;
  const int n_qt = (S + 63) / 64;
  const int span = n_qt - qt_start;
  const int total = rep * span;

  // stage iteration `it`'s tiles + lse/dr into buffer `buf`.  The
  // lse/dr fill runs with NO glds outstanding (callers order it right
  // after the barrier), so hipcc's ordinary-load waits stay counted.
  auto fill_lse = [&](int it, char* buf) {
    const int g = it / span;
    const int qt = qt_start + it % span;
    const int h = kvh * rep + g;
    const long roff = (long)(b * H + h) * S;
    const int q0 = qt * 64;
    float* lse_b = (float*)(buf + 2 * KVB * D * 2 + 2 * D * KVB * 2);
    float* dr_b = lse_b + KVB;
    for (int i = tid; i < KVB; i += NTHREADS) {
      int qr = q0 + i;
      lse_b[i] = qr < S ? LSE[roff + qr] : 3e30f;
      dr_b[i] = qr < S ? DR[roff + qr] : 0.f;
    }
  };
  auto stage = [&](int it, char* buf) {
    const int g = it / span;
    const int qt = qt_start + it % span;
    const int h = kvh * rep + g;
    const long qvb = (long)b * qsB + (long)h * qsH;
    const long dob = (long)b * osB + (long)h * osH;
    const long offT = ((long)(b * H + h) * D) * S;
    const int q0 = qt * 64;
    char* q_b = buf;
    char* do_b = q_b + KVB * D * 2;
    char* qt_b = do_b + KVB * D * 2;
    char* dot_b = qt_b + D * KVB * 2;
    glds_tile<KVB, RB>(Q + qvb, qsS, q0, S, 0, D, q_b, wid, lane);
    glds_tile<KVB, RB>(dO + dob, osS, q0, S, 0, D, do_b, wid, lane);
    glds_tile<D, KVB * 2>(Qt + offT, S, 0, D, q0, S, qt_b, wid, lane);
    glds_tile<D, KVB * 2>(dOt + offT, S, 0, D, q0, S, dot_b, wid, lane);
  };

  fill_lse(0, lds);
  stage(0, lds);

  for (int it = 0; it < total; ++it) {
    char* cur = lds + (it & 1) * STG;
    char* nxt = lds + ((it + 1) & 1) * STG;
    // drains cur's DMA (issued a full iteration ago) + separates the
    // previous iteration's reads of nxt's buffer from the overwrite
    __syncthreads();
    if (it + 1 < total) {
      fill_lse(it + 1, nxt);
      stage(it + 1, nxt);
    }
    char* q_lds = cur;
    char* do_lds = q_lds + KVB * D * 2;
    char* qt_lds = do_lds + KVB * D * 2;
    char* dot_lds = qt_lds + D * KVB * 2;
    float* lse_lds = (float*)(dot_lds + D * KVB * 2);
    float* dr_lds = lse_lds + KVB;
    const int g = it / span;
    const int qt = qt_start + it % span;
    const int q0 = qt * 64;
    (void)g;

    float4v st[4], dpt[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      st[f] = {0.f, 0.f, 0.f, 0.f};
      dpt[f] = {0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int kk = 0; kk < KC; ++kk) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 qf = *reinterpret_cast<const bf16x8*>(
            q_lds + swz(f * 16 + (lane & 15),
                        kk * 64 + (lane >> 4) * 16, RB));
        st[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            kreg[kk], qf, st[f], 0, 0, 0);
        bf16x8 dof = *reinterpret_cast<const bf16x8*>(
            do_lds + swz(f * 16 + (lane & 15),
                         kk * 64 + (lane >> 4) * 16, RB));
        dpt[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            vreg[kk], dof, dpt[f], 0, 0, 0);
      }
    }
    const int krow0 = kv0 + wid * 16 + (lane >> 4) * 4;
    // P^T -> wave-private LDS, dV += P^T dO
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int qcol = q0 + f * 16 + (lane & 15);
      float lse_q = lse_lds[f * 16 + (lane & 15)];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        bool dead = qcol >= S || lse_q > 1e30f;
        if (CAUSAL) dead |= qcol < krow0 + r;
        float p = dead ? 0.f : __expf(st[f][r] - lse_q);
        st[f][r] = p;  // keep P for the dS pass below
        *reinterpret_cast<short*>(
            my_p + swz((lane >> 4) * 4 + r, (f * 16 + (lane & 15)) * 2,
                       KVB * 2)) = f32_to_bf16(p);
      }
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 pf = *reinterpret_cast<const bf16x8*>(
          my_p + swz(lane & 15, kk * 64 + (lane >> 4) * 16, KVB * 2));
#pragma unroll
      for (int f = 0; f < DF; ++f) {
        bf16x8 dotf = *reinterpret_cast<const bf16x8*>(
            dot_lds + swz(f * 16 + (lane & 15),
                          kk * 64 + (lane >> 4) * 16, KVB * 2));
        dv_acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pf, dotf, dv_acc[f], 0, 0, 0);
      }
    }
    // dS^T over the SAME wave-private buffer, dK += dS^T Q
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      float dr_q = dr_lds[f * 16 + (lane & 15)];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float ds = st[f][r] * (dpt[f][r] - dr_q) * scale;
        *reinterpret_cast<short*>(
            my_p + swz((lane >> 4) * 4 + r, (f * 16 + (lane & 15)) * 2,
                       KVB * 2)) = f32_to_bf16(ds);
      }
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 dsf = *reinterpret_cast<const bf16x8*>(
          my_p + swz(lane & 15, kk * 64 + (lane >> 4) * 16, KVB * 2));
#pragma unroll
      for (int f = 0; f < DF; ++f) {
        bf16x8 qtf = *reinterpret_cast<const bf16x8*>(
            qt_lds + swz(f * 16 + (lane & 15),
                         kk * 64 + (lane >> 4) * 16, KVB * 2));
        dk_acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsf, qtf, dk_acc[f], 0, 0, 0);
      }
    }
  }

  const int orow0 = kv0 + wid * 16 + (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = orow0 + r;
    if (row >= S) continue;
    short* dkr = dK + kvb + (long)row * qsS;
    short* dvr = dV + kvb + (long)row * qsS;
#pragma unroll
    for (int f = 0; f < DF; ++f) {
      dkr[f * 16 + (lane & 15)] = f32_to_bf16(dk_acc[f][r]);
      dvr[f * 16 + (lane & 15)] = f32_to_bf16(dv_acc[f][r]);
    }
  }
}

template <int D, bool CAUSAL, bool STRIDED>
__global__ __launch_bounds__(NTHREADS, 2) void attn_bwd_q_kernel(
    const short* __restrict__ Q, const short* __restrict__ K,
    const short* __restrict__ V, const short* __restrict__ dO,
    const short* __restrict__ Kt, const float* __restrict__ LSE,
    const float* __restrict__ DR, short* __restrict__ dQ,
    int B, int H, int HK, int S, float scale,
    long qsB, long qsH, long qsS_, long osB, long osH, long osS_) {
  const long qsS = STRIDED ? qsS_ : (long)D;
  const long osS = STRIDED ? osS_ : (long)D;
  constexpr int DF = D / 16;
  constexpr int KC = D / 32;
  constexpr int RB = D * 2;
  __shared__ __attribute__((aligned(16))) char lds[
      2 * KVB * D * 2 +      // K, V [64][D]
      D * KVB * 2 +          // Kt [D][64]
      4 * 16 * KVB * 2];     // per-wave dS [16][64]
  char* k_lds = lds;
  char* v_lds = k_lds + KVB * D * 2;
  char* kt_lds = v_lds + KVB * D * 2;
  char* p_base = kt_lds + D * KVB * 2;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  char* my_ds = p_base + wid * 16 * KVB * 2;

  const int n_qt = (S + 63) / 64;
  const int bh = blockIdx.x / n_qt;
  const int qt = blockIdx.x % n_qt;
  const int b = bh / H, h = bh % H;
  const int q0 = qt * 64 + wid * 16;

  const long qvb = (long)b * qsB + (long)h * qsH;
  const long dob = (long)b * osB + (long)h * osH;
  const int kh = h / (H / HK);                    // GQA kv head
  const long kvb = STRIDED ? qvb
      : ((long)(b * HK + kh) * S) * (long)D;      // K/V base
  const long offT = ((long)(b * HK + kh) * D) * S;  // Kt
  const long roff = (long)(b * H + h) * S;

  // Q (scaled) and dO fragments
  bf16x8 qreg[KC], doreg[KC];
  {
    int row = q0 + (lane & 15);
    if (row >= S) row = S - 1;
#pragma unroll
    for (int c = 0; c < KC; ++c) {
      int d0 = c * 32 + (lane >> 4) * 8;
      bf16x8 v8 = *reinterpret_cast<const bf16x8*>(
          Q + qvb + (long)row * qsS + d0);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        v8[j] = (__bf16)((float)v8[j] * scale);
      qreg[c] = v8;
      doreg[c] = *reinterpret_cast<const bf16x8*>(
          dO + dob + (long)row * osS + d0);
    }
  }
  // per-lane lse/Drow for this lane's 4 C-rows
  float lse_r[4], dr_r[4];
  {
    int r0 = qt * 64 + wid * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int row = r0 + r;
      lse_r[r] = row < S ? LSE[roff + row] : 3e30f;
      dr_r[r] = row < S ? DR[roff + row] : 0.f;
    }
  }

  float4v dq_acc[DF];
#pragma unroll
  for (int f = 0; f < DF; ++f) dq_acc[f] = {0.f, 0.f, 0.f, 0.f};

  const int q_tile_end = qt * 64 + 63;
  const int n_kv = CAUSAL
      ? min((q_tile_end / KVB) + 1, (S + KVB - 1) / KVB)
      : (S + KVB - 1) / KVB;
  for (int t = 0; t < n_kv; ++t) {
    const int kv0 = t * KVB;
    glds_tile<KVB, RB>(K + kvb, qsS, kv0, S, 0, D, k_lds, wid, lane);
    glds_tile<KVB, RB>(V + kvb, qsS, kv0, S, 0, D, v_lds, wid, lane);
    glds_tile<D, KVB * 2>(Kt + offT, S, 0, D, kv0, S, kt_lds, wid, lane);
    __syncthreads();

    float4v st[4], dpt[4];
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      st[f] = {0.f, 0.f, 0.f, 0.f};
      dpt[f] = {0.f, 0.f, 0.f, 0.f};
    }
#pragma unroll
    for (int kk = 0; kk < KC; ++kk) {
#pragma unroll
      for (int f = 0; f < 4; ++f) {
        bf16x8 kf = *reinterpret_cast<const bf16x8*>(
            k_lds + swz(f * 16 + (lane & 15),
                        kk * 64 + (lane >> 4) * 16, RB));
        st[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qreg[kk], kf, st[f], 0, 0, 0);
        bf16x8 vf = *reinterpret_cast<const bf16x8*>(
            v_lds + swz(f * 16 + (lane & 15),
                        kk * 64 + (lane >> 4) * 16, RB));
        dpt[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            doreg[kk], vf, dpt[f], 0, 0, 0);
      }
    }
    const int lrow0 = qt * 64 + wid * 16 + (lane >> 4) * 4;
#pragma unroll
    for (int f = 0; f < 4; ++f) {
      int kcol = kv0 + f * 16 + (lane & 15);
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        bool dead = kcol >= S || lse_r[r] > 1e30f;
        if (CAUSAL) dead |= kcol > lrow0 + r;
        float p = dead ? 0.f : __expf(st[f][r] - lse_r[r]);
        float ds = p * (dpt[f][r] - dr_r[r]) * scale;
        int row = (lane >> 4) * 4 + r;
        int cb = (f * 16 + (lane & 15)) * 2;
        *reinterpret_cast<short*>(
            my_ds + swz(row, cb, KVB * 2)) = f32_to_bf16(ds);
      }
    }
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {
      bf16x8 dsf = *reinterpret_cast<const bf16x8*>(
          my_ds + swz(lane & 15, kk * 64 + (lane >> 4) * 16, KVB * 2));
#pragma unroll
      for (int f = 0; f < DF; ++f) {
        bf16x8 ktf = *reinterpret_cast<const bf16x8*>(
            kt_lds + swz(f * 16 + (lane & 15),
                         kk * 64 + (lane >> 4) * 16, KVB * 2));
        dq_acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsf, ktf, dq_acc[f], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  const int orow0 = qt * 64 + wid * 16 + (lane >> 4) * 4;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int row = orow0 + r;
    if (row >= S) continue;
    short* dqr = dQ + qvb + (long)row * qsS;
#pragma unroll
    for (int f = 0; f < DF; ++f)
      dqr[f * 16 + (lane & 15)] = f32_to_bf16(dq_acc[f][r]);
  }
}

}  // namespace

torch::Tensor transpose2d(torch::Tensor x);  // gemm_mfma.hip (batched)
torch::Tensor transpose2d_bhsd(torch::Tensor x);

namespace {
// q/k/v must be contiguous OR share one (b,h,s) stride triple with
// unit d-stride and 16-byte-aligned rows — exactly what slices of a
// packed [B,S,3,H,D] QKV projection satisfy.
void check_qkv_strides(const torch::Tensor& q, const torch::Tensor& k,
                       const torch::Tensor& v) {
  TORCH_CHECK(q.stride(3) == 1 && k.strides() == q.strides() &&
              v.strides() == q.strides(),
              "attn: q/k/v must share one stride layout (d contiguous)");
  TORCH_CHECK(q.stride(2) % 8 == 0 && q.stride(1) % 8 == 0 &&
              q.stride(0) % 8 == 0,
              "attn: strides must keep rows 16B-aligned");
}
}  // namespace

std::vector<torch::Tensor> attn_fwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, bool causal) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(q.dim() == 4, "attn_fwd: [B,H,S,D]");
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int HK = k.size(1);
  TORCH_CHECK(k.sizes() == v.sizes() && k.size(0) == B &&
              k.size(2) == S && k.size(3) == D && H % HK == 0,
              "attn_fwd: kv heads must divide q heads (GQA)");
  TORCH_CHECK(D == 64 || D == 128, "attn_fwd: D must be 64 or 128");
  TORCH_CHECK(S % 8 == 0, "attn_fwd: S must be a multiple of 8");
  const bool cont = q.is_contiguous() && k.is_contiguous() &&
                    v.is_contiguous();
  TORCH_CHECK(cont || HK == H,
              "attn_fwd: GQA (kv heads < heads) needs contiguous q/k/v");
  if (!cont) check_qkv_strides(q, k, v);
  // pre-transpose V: [B,H,S,D] -> [B,H,D,S] (one memory-bound pass on
  // the batched LDS-tiled transpose; lets the PV B-fragments read
  // 8 bf16 contiguous along kv)
  auto vt = cont ? transpose2d(v.reshape({(long)B * HK, S, D}))
                       .reshape({B, HK, D, S})
                 : transpose2d_bhsd(v);
  // strided (packed-QKV) path: O goes out as a [B,S,H,D] buffer viewed
  // [B,H,S,D], so the caller's head-merge reshape is a free view
  auto o = cont ? torch::empty({B, H, S, D}, q.options())
                : torch::empty({B, S, H, D}, q.options())
                      .permute({0, 2, 1, 3});
  auto lse = torch::empty({B, H, S},
                          q.options().dtype(torch::kFloat32));
  float scale = 1.0f / sqrtf((float)D);
  // 8-wave blocks (128 q rows) halve per-FLOP K/V re-reads at long S;
  // 4-wave blocks keep the grid full at short S.
  static const int force = [] {
    const char* e = getenv("GSAMD_ATTN_NW");
    return e ? atoi(e) : 0;
  }();
  // A/B (within-box, repeated): 8-wave wins at S=1024-2048 (+6%),
  // loses at S=4096 (-8%); gate to the measured window.
  bool wide = (long)B * H * ((S + 127) / 128) >= 256 && S >= 1024 &&
              S <= 2048;
  if (force == 4) wide = false;
  if (force == 8) wide = true;
  const int QB = wide ? 128 : 64;
  int grid = B * H * ((S + QB - 1) / QB);
  auto stream = c10::cuda::getCurrentCUDAStream();
#define LAUNCH(DD, CC, NW)                                               \
  hipLaunchKernelGGL((attn_fwd_kernel<DD, CC, NW, SR>), dim3(grid),          \
                     dim3(NW * 64), 0, stream.stream(),                  \
                     (const short*)q.data_ptr(),                         \
                     (const short*)k.data_ptr(),                         \
                     (const short*)vt.data_ptr(),                        \
                     (short*)o.data_ptr(), lse.data_ptr<float>(),        \
                     B, H, HK, S, scale, q.stride(0), q.stride(1),       \
                     q.stride(2), o.stride(0), o.stride(1), o.stride(2))
#define LAUNCH_D(DD, SR)                                                 \
  do {                                                                   \
    if (causal) { if (wide) LAUNCH(DD, true, 8);                         \
                  else LAUNCH(DD, true, 4); }                            \
    else { if (wide) LAUNCH(DD, false, 8);                               \
           else LAUNCH(DD, false, 4); }                                  \
  } while (0)
  if (cont) {
    if (D == 64) { constexpr bool SR = false; LAUNCH_D(64, SR); }
    else { constexpr bool SR = false; LAUNCH_D(128, SR); }
  } else {
    if (D == 64) { constexpr bool SR = true; LAUNCH_D(64, SR); }
    else { constexpr bool SR = true; LAUNCH_D(128, SR); }
  }
#undef LAUNCH_D
#undef LAUNCH
  HIP_CHECK_LAST();
  return {o, lse};
}

// Contiguous q/k/v: returns {dq, dk, dv} (each contiguous).  Strided
// (packed-QKV) q/k/v: allocates ONE [B,S,3,H,D] buffer, writes the
// three grads into it as strided views, and returns
// {dq, dk, dv, dqkv} — the caller hands dqkv straight back as the
// packed projection's grad (no select-backward zeros/copies, no cat).
std::vector<torch::Tensor> attn_bwd(torch::Tensor q, torch::Tensor k,
                                    torch::Tensor v, torch::Tensor o,
                                    torch::Tensor dout,
                                    torch::Tensor lse, bool causal) {
  const int B = q.size(0), H = q.size(1), S = q.size(2), D = q.size(3);
  const int HK = k.size(1);
  const bool cont = q.is_contiguous() && k.is_contiguous() &&
                    v.is_contiguous();
  TORCH_CHECK(cont || HK == H,
              "attn_bwd: GQA (kv heads < heads) needs contiguous q/k/v");
  if (!cont) check_qkv_strides(q, k, v);
  TORCH_CHECK(o.stride(3) == 1 && dout.strides() == o.strides(),
              "attn_bwd: dout must match o's layout");
  auto stream = c10::cuda::getCurrentCUDAStream();
  // pre-transposes (batched LDS-tiled kernel) + Drow
  auto qt = cont ? transpose2d(q.reshape({(long)B * H, S, D}))
                 : transpose2d_bhsd(q);
  auto kt = cont ? transpose2d(k.reshape({(long)B * HK, S, D}))
                 : transpose2d_bhsd(k);
  auto dot = dout.is_contiguous()
                 ? transpose2d(dout.reshape({(long)B * H, S, D}))
                 : transpose2d_bhsd(dout);
  long n_rows = (long)B * H * S;
  auto drow = torch::empty({n_rows}, q.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL(attn_drow_kernel,
                     dim3((int)((n_rows + 3) / 4)), dim3(256), 0,
                     stream.stream(), (const short*)dout.data_ptr(),
                     (const short*)o.data_ptr(), drow.data_ptr<float>(),
                     n_rows, H, S, D, o.stride(0), o.stride(1),
                     o.stride(2));
  torch::Tensor dq, dk, dv, dqkv;
  if (cont) {
    dq = torch::empty({B, H, S, D}, q.options());
    dk = torch::empty({B, HK, S, D}, q.options());
    dv = torch::empty({B, HK, S, D}, q.options());
  } else {
    dqkv = torch::empty({B, S, 3, H, D}, q.options());
    dq = dqkv.select(2, 0).permute({0, 2, 1, 3});
    dk = dqkv.select(2, 1).permute({0, 2, 1, 3});
    dv = dqkv.select(2, 2).permute({0, 2, 1, 3});
    TORCH_CHECK(dq.strides() == q.strides(),
                "attn_bwd: packed grads must mirror the input layout");
  }
  float scale = 1.0f / sqrtf((float)D);
  int grid_kv = B * HK * ((S + KVB - 1) / KVB);
  int grid_q = B * H * ((S + 63) / 64);
  static const bool pipe = [] {
    // D=128 pipelined dK/dV variant (double-buffered staging, dS
    // aliased onto P, one barrier/iter).  MEASURED: wash vs the base
    // kernel (llama-8B 183.8 vs 183.1 ms/iter) — at 1 wave/SIMD the
    // iteration is not drain-bound.  Kept gated.
    const char* e = getenv("GSAMD_ATTN_PIPE");
    return e && e[0] == '1';
  }();
#define LAUNCH_KV(DD, CC)                                               \
  do {                                                                  \
    if (DD == 128 && pipe)                                              \
      hipLaunchKernelGGL((attn_bwd_kv_pipe_kernel<DD, CC, SR>),         \
                         dim3(grid_kv), dim3(NTHREADS), 0,              \
                         stream.stream(),                               \
                         (const short*)q.data_ptr(),                    \
                         (const short*)k.data_ptr(),                    \
                         (const short*)v.data_ptr(),                    \
                         (const short*)dout.data_ptr(),                 \
                         (const short*)qt.data_ptr(),                   \
                         (const short*)dot.data_ptr(),                  \
                         lse.data_ptr<float>(), drow.data_ptr<float>(), \
                         (short*)dk.data_ptr(), (short*)dv.data_ptr(),  \
                         B, H, HK, S, scale, q.stride(0), q.stride(1),  \
                         q.stride(2), o.stride(0), o.stride(1),         \
                         o.stride(2));                                  \
    else                                                                \
      hipLaunchKernelGGL((attn_bwd_kv_kernel<DD, CC, SR>),              \
                         dim3(grid_kv), dim3(NTHREADS), 0,              \
                         stream.stream(),                               \
                         (const short*)q.data_ptr(),                    \
                         (const short*)k.data_ptr(),                    \
                         (const short*)v.data_ptr(),                    \
                         (const short*)dout.data_ptr(),                 \
                         (const short*)qt.data_ptr(),                   \
                         (const short*)dot.data_ptr(),                  \
                         lse.data_ptr<float>(), drow.data_ptr<float>(), \
                         (short*)dk.data_ptr(), (short*)dv.data_ptr(),  \
                         B, H, HK, S, scale, q.stride(0), q.stride(1),  \
                         q.stride(2), o.stride(0), o.stride(1),         \
                         o.stride(2));                                  \
  } while (0)
#define LAUNCH_BWD(DD, CC)                                              \
  do {                                                                  \
    LAUNCH_KV(DD, CC);                                                  \
    hipLaunchKernelGGL((attn_bwd_q_kernel<DD, CC, SR>), dim3(grid_q),   \
                       dim3(NTHREADS), 0, stream.stream(),              \
                       (const short*)q.data_ptr(),                      \
                       (const short*)k.data_ptr(),                      \
                       (const short*)v.data_ptr(),                      \
                       (const short*)dout.data_ptr(),                   \
                       (const short*)kt.data_ptr(),                     \
                       lse.data_ptr<float>(), drow.data_ptr<float>(),   \
                       (short*)dq.data_ptr(), B, H, HK, S, scale,       \
                       q.stride(0), q.stride(1), q.stride(2),           \
                       o.stride(0), o.stride(1), o.stride(2));          \
  } while (0)
  if (cont) {
    constexpr bool SR = false;
    if (D == 64) {
      if (causal) LAUNCH_BWD(64, true); else LAUNCH_BWD(64, false);
    } else {
      if (causal) LAUNCH_BWD(128, true); else LAUNCH_BWD(128, false);
    }
  } else {
    constexpr bool SR = true;
    if (D == 64) {
      if (causal) LAUNCH_BWD(64, true); else LAUNCH_BWD(64, false);
    } else {
      if (causal) LAUNCH_BWD(128, true); else LAUNCH_BWD(128, false);
    }
  }
#undef LAUNCH_BWD
  HIP_CHECK_LAST();
  if (cont) return {dq, dk, dv};
  return {dq, dk, dv, dqkv};
}
