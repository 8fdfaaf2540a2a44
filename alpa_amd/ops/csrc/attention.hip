// Flash attention forward (bf16, causal/full) for gfx950 — MFMA
// 16x16x32_bf16, online softmax, never materializes S x S.
//
// Covers the reference's fused-attention slot (SURVEY.md §2.3 kernel table:
// "batched GEMM + softmax ... fused attention kernel (QK^T->softmax->V)").
//
// Structure: 256 threads = 4 waves per block; block owns 64 q rows of one
// (batch, head); wave owns 16 rows.  K tile [64][Dp] and transposed V tile
// [Dp][64] staged in LDS, shared by all waves; per-wave P tile round-trips
// through LDS to re-fragment S (C-layout) into the PV A-operand.
//
// MFMA fragment layouts (verified on hardware by the mfma_probe test):
//   A (16x32): m = lane&15, k = (lane>>4)*8 + j   (j = 0..7)
//   B (32x16): n = lane&15, k = (lane>>4)*8 + j
//   C/D      : n = lane&15, m = (lane>>4)*4 + reg (f32x4)
#include "common.h"
#include <cstdlib>

#define ATTN_BLOCK_Q 64
#define ATTN_BLOCK_K 64
#define ATTN_THREADS 1024
#define ATTN_BWD_THREADS 512

// Strides are in elements; the innermost (D) dim must be contiguous.
// Strided addressing lets the packed qkv layout [B, S, heads, 3*D] feed the
// kernel directly — no permute/contiguous copies on the hot path.
struct AttnStrides {
  int64_t qb, qh, qs;  // q batch/head/seq strides
  int64_t kb, kh, ks;
  int64_t vb, vh, vs;
  int64_t ob, oh, os;
};

// XCD-aware workgroup swizzle (gfx950: 8 XCDs, hardware assigns
// workgroup i to XCD i%8, each XCD has its own L2): remap so each XCD
// gets a CONTIGUOUS run of logical blocks — all blocks of one
// (batch,head) then share one L2 and its K/V (fwd) / Q,dO (bwd) tiles
// are loaded from HBM once per XCD instead of once per block.
__device__ inline int xcd_swizzle(int flat, int total) {
  constexpr int NXCD = 8;
  if (total % NXCD != 0) return flat;
  return (flat % NXCD) * (total / NXCD) + flat / NXCD;
}

// ABL: ablation level for perf diagnosis (0 = full kernel; higher skips
// later phases; asm keep-alives prevent dead-code elimination of earlier
// phases — guide methodology rule 17)
template <int Dp, int ABL = 0, bool AL = false, bool VL = false,
          bool DB = true, int NT = ATTN_THREADS>
__global__ __launch_bounds__(NT) void attn_fwd_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ o,
    float* __restrict__ lse_out, int H, int S, int Skv, int D, float scale,
    int causal, const float* __restrict__ alibi,
    const int* __restrict__ kv_lens, AttnStrides st) {
  // 8 waves x 16 q-rows = 128 q rows per block; K/V tiles of 64 kv are
  // double-buffered in LDS with register-prefetched staging (loads for
  // tile t+1 issue before computing tile t and land after it — the HBM
  // latency hides under the MFMA/softmax work), one barrier per tile.
  constexpr int KSTEPS_QK = Dp / 32;
  constexpr int NTILES = ATTN_BLOCK_K / 16;  // 4
  constexpr int DTILES = Dp / 16;
  constexpr int LP = 4;  // padding: (Dp+LP)*2B stride -> gcd(words,32)=2, 16 distinct banks (LP=8 gave gcd 4 => 2-way conflicts)
  constexpr int NW = NT / 64;                // waves per block
  constexpr int GPR = Dp / 8;                // bf16x8 groups per kv row
  constexpr int TOTAL_G = ATTN_BLOCK_K * GPR;
  constexpr int G_PER_T = (TOTAL_G + NT - 1) / NT;

  // XCD co-location + causal ordering: swizzle the flattened id so one
  // XCD owns contiguous (bh, qb) work; within a bh, later q blocks
  // (more kv tiles under causal) go first to avoid a heavy tail
  const int flat_ = xcd_swizzle(
      (int)(blockIdx.y * gridDim.x + blockIdx.x),
      (int)(gridDim.x * gridDim.y));
  const int qb_raw = flat_ % (int)gridDim.x;
  const int qb = causal ? ((int)gridDim.x - 1 - qb_raw) : qb_raw;
  const int bh = flat_ / (int)gridDim.x;
  const int batch = bh / H, head = bh % H;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lo = lane & 15;
  const int hi = lane >> 4;

  const short* qp = q + batch * st.qb + head * st.qh;
  const short* kp = k + batch * st.kb + head * st.kh;
  const short* vp = v + batch * st.vb + head * st.vh;
  short* op = o + batch * st.ob + head * st.oh;
  const int q_row0 = qb * (16 * NW) + wave * 16;

  // DB=false: single-buffered K/V (64 KB total for Dp=96 instead of
  // 91 KB) fits TWO blocks per CU — inter-block latency hiding replaces
  // the intra-block prefetch (experiment; see profiles/ PMC notes)
  __shared__ short k_lds[DB ? 2 : 1][ATTN_BLOCK_K][Dp + LP];
  __shared__ short vt_lds[DB ? 2 : 1][Dp][ATTN_BLOCK_K + LP];
  __shared__ short p_lds[NW][16][ATTN_BLOCK_K + LP];

  // ---- Q fragments in registers ----
  bf16x8 q_frag[KSTEPS_QK];
  {
    int row = min(q_row0 + lo, S - 1);
#pragma unroll
    for (int ks = 0; ks < KSTEPS_QK; ++ks) {
      int col = ks * 32 + hi * 8;
      bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
      q_frag[ks] = z;
      if (col + 8 <= D)
        q_frag[ks] =
            *reinterpret_cast<const bf16x8*>(qp + (int64_t)row * st.qs + col);
    }
  }

  // Swapped-operand form: the wave computes S^T = K Q^T, so each lane
  // holds 16 S values of ONE q column (q = q_row0 + lo) — the softmax
  // row-reduction is 16 in-register ops + 2 shfl steps across the 4
  // hi-groups (vs 8 serial shfl chains per row in the naive layout).
  // PV becomes O^T = V^T P^T with both operands read contiguously.
  float m_state = -INFINITY, l_state = 0.f;
  f32x4 o_acc[DTILES];  // O^T C-layout: d = hi*4+r (+16*dt), q = lo
#pragma unroll
  for (int dt = 0; dt < DTILES; ++dt) o_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kv_limit =
      causal ? min(Skv, qb * (16 * NW) + 16 * NW) : Skv;
  const int n_tiles = (kv_limit + ATTN_BLOCK_K - 1) / ATTN_BLOCK_K;
  const int my_q = q_row0 + lo;  // this lane's q row
  // ALiBi (BLOOM): per-head slope, bias = slope * (kv_pos - q_pos); the
  // q GLOBAL position is my_q + (Skv - S) for cached decode
  // varlen (continuous batching): this batch slot's real kv length;
  // tiles past it are masked per element (Skv stays the tile loop bound)
  const int my_skv = VL ? kv_lens[batch] : Skv;
  const float al_slope = AL ? alibi[bh % H] : 0.f;
  const int al_qoff = my_skv - S;

  // staging: load tile -> regs (two kv rows per thread so the V transpose
  // writes pair as b32)
  bf16x8 kreg[2 * G_PER_T], vreg[2 * G_PER_T];
  auto issue_loads = [&](int tile) {
#pragma unroll
    for (int it = 0; it < G_PER_T; ++it) {
      int t = threadIdx.x + it * NT;
      bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
      kreg[2 * it] = z; kreg[2 * it + 1] = z;
      vreg[2 * it] = z; vreg[2 * it + 1] = z;
      if (t < TOTAL_G / 2) {
        int kvr = (t / GPR) * 2, dg = (t % GPR) * 8;
#pragma unroll
        for (int u = 0; u < 2; ++u) {
          int src = tile * ATTN_BLOCK_K + kvr + u;
          if (src < Skv && dg + 8 <= D) {
            kreg[2 * it + u] = *reinterpret_cast<const bf16x8*>(
                kp + (int64_t)src * st.ks + dg);
            vreg[2 * it + u] = *reinterpret_cast<const bf16x8*>(
                vp + (int64_t)src * st.vs + dg);
          }
        }
      }
    }
  };
  auto write_tile = [&](int buf) {
#pragma unroll
    for (int it = 0; it < G_PER_T; ++it) {
      int t = threadIdx.x + it * NT;
      if (t < TOTAL_G / 2) {
        int kvr = (t / GPR) * 2, dg = (t % GPR) * 8;
        *reinterpret_cast<bf16x8*>(&k_lds[buf][kvr][dg]) = kreg[2 * it];
        *reinterpret_cast<bf16x8*>(&k_lds[buf][kvr + 1][dg]) =
            kreg[2 * it + 1];
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          short2 pr;
          pr.x = vreg[2 * it][j];
          pr.y = vreg[2 * it + 1][j];
          *reinterpret_cast<short2*>(&vt_lds[buf][dg + j][kvr]) = pr;
        }
      }
    }
  };

  auto advance = [&](int ti, int cur) {
    if (ti + 1 >= n_tiles) return;
    if (DB) {
      write_tile(1 - cur);  // waits on the prefetched loads here
      __syncthreads();
    } else {
      __syncthreads();      // everyone done reading buf 0
      issue_loads(ti + 1);
      write_tile(0);
      __syncthreads();
    }
  };

  if (n_tiles > 0) {
    issue_loads(0);
    write_tile(0);
    __syncthreads();
  }

  for (int ti = 0; ti < n_tiles; ++ti) {
    const int kvb = ti * ATTN_BLOCK_K;
    const int cur = DB ? (ti & 1) : 0;
    if (DB && ti + 1 < n_tiles) issue_loads(ti + 1);  // lands under compute

    // causal: tiles entirely above this wave's q rows contribute nothing
    if (causal && kvb > q_row0 + 15) {
      advance(ti, cur);
      continue;
    }

    if (ABL >= 4) {  // staging-only
      float keep = bf2f(k_lds[cur][lane][0]) + bf2f(vt_lds[cur][lane][0]);
      asm volatile("" ::"v"(keep));
      advance(ti, cur);
      continue;
    }

__builtin_amdgcn_s_setprio(1);  // T5: favor the MFMA cluster
    // ---- S^T = K Q^T (A = K tile from LDS, B = Q fragments) ----
    f32x4 s_acc[NTILES];
#pragma unroll
    for (int nt = 0; nt < NTILES; ++nt) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < KSTEPS_QK; ++ks) {
        bf16x8 a = *reinterpret_cast<const bf16x8*>(
            &k_lds[cur][nt * 16 + lo][ks * 32 + hi * 8]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, q_frag[ks], acc, 0,
                                                      0, 0);
      }
      s_acc[nt] = acc;
    }
    __builtin_amdgcn_s_setprio(0);

    if (ABL >= 3) {  // QK^T only
#pragma unroll
      for (int nt = 0; nt < NTILES; ++nt)
        asm volatile("" ::"v"(s_acc[nt][0]), "v"(s_acc[nt][3]));
      advance(ti, cur);
      continue;
    }
    if (ABL == 2) {  // no softmax math: raw S write + one read
#pragma unroll
      for (int nt = 0; nt < NTILES; ++nt)
#pragma unroll
        for (int r = 0; r < 4; ++r)
          p_lds[wave][lo][nt * 16 + hi * 4 + r] = f2bf(s_acc[nt][r]);
      bf16x8 a0 = *reinterpret_cast<const bf16x8*>(&p_lds[wave][lo][hi * 8]);
      asm volatile("" ::"v"(a0));
      advance(ti, cur);
      continue;
    }

    // ---- mask + scale + lane-local max over this lane's 16 kv ----
    float tile_max = -INFINITY;
#pragma unroll
    for (int nt = 0; nt < NTILES; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int kv_idx = kvb + nt * 16 + hi * 4 + r;
        float sv = s_acc[nt][r] * scale;
        if (AL)  // compile-time: the non-ALiBi instantiation is untouched
          sv = fmaf(al_slope, (float)(kv_idx - my_q - al_qoff), sv);
        bool masked = (kv_idx >= my_skv) || (causal && kv_idx > my_q) ||
                      (my_q >= S);
        sv = masked ? -INFINITY : sv;
        s_acc[nt][r] = sv;
        tile_max = fmaxf(tile_max, sv);
      }
    }
    // cross-hi reduce (lanes lo, lo+16, lo+32, lo+48 share q column)
    tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 16, 64));
    tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));

    // ---- online rescale; P^T = exp(S^T - m); sum ----
    // T13 defer-max: skip the O-wide rescale while the max grows by
    // less than THR (P values then bounded by e^THR=2981, fine in f32
    // accum); only when some lane's max jumps past the threshold does
    // the wave rescale and advance m (guide T13; isolated +5%).
    constexpr float DEFER_THR = 8.0f;
    if (!__all(tile_max <= m_state + DEFER_THR)) {
      float m_new = fmaxf(m_state, tile_max);
      float alpha = (m_state == -INFINITY) ? 0.f : __expf(m_state - m_new);
      m_state = m_new;
      l_state *= alpha;
#pragma unroll
      for (int dt = 0; dt < DTILES; ++dt) {
        o_acc[dt][0] *= alpha; o_acc[dt][1] *= alpha;
        o_acc[dt][2] *= alpha; o_acc[dt][3] *= alpha;
      }
    }
    float part = 0.f;
#pragma unroll
    for (int nt = 0; nt < NTILES; ++nt) {
      bf16x4 pk;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float pval = (s_acc[nt][r] == -INFINITY)
                         ? 0.f
                         : __expf(s_acc[nt][r] - m_state);
        part += pval;
        pk[r] = f2bf(pval);
      }
      // P^T transposed store: p_lds[q][kv], 4 consecutive kv per lane
      // packed into ONE ds_write_b64 (was 4 b16 writes)
      *reinterpret_cast<bf16x4*>(&p_lds[wave][lo][nt * 16 + hi * 4]) = pk;
    }
    part += __shfl_xor(part, 16, 64);
    part += __shfl_xor(part, 32, 64);
    l_state += part;

    if (ABL >= 1) {
      asm volatile("" ::"v"(l_state), "v"(m_state));
      advance(ti, cur);
      continue;
    }

__builtin_amdgcn_s_setprio(1);  // T5: favor the MFMA cluster
    // ---- O^T += V^T P^T  (A = V^T from vt_lds, B = P^T from p_lds) ----
#pragma unroll
    for (int ks = 0; ks < ATTN_BLOCK_K / 32; ++ks) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &p_lds[wave][lo][ks * 32 + hi * 8]);
#pragma unroll
      for (int dt = 0; dt < DTILES; ++dt) {
        bf16x8 a = *reinterpret_cast<const bf16x8*>(
            &vt_lds[cur][dt * 16 + lo][ks * 32 + hi * 8]);
        o_acc[dt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, o_acc[dt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    advance(ti, cur);
  }

  // ---- epilogue: lane holds O^T[d = 16*dt + hi*4 + r][q = lo] ----
  {
    int q_idx = q_row0 + lo;
    if (q_idx < S) {
      float inv_l = (l_state > 0.f) ? 1.0f / l_state : 0.f;
#pragma unroll
      for (int dt = 0; dt < DTILES; ++dt) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          int col = dt * 16 + hi * 4 + r;
          if (col < D)
            op[(int64_t)q_idx * st.os + col] = f2bf(o_acc[dt][r] * inv_l);
        }
      }
      if (hi == 0 && lse_out != nullptr)
        lse_out[(int64_t)bh * S + q_idx] =
            (l_state > 0.f) ? m_state + __logf(l_state) : -INFINITY;
    }
  }
}

// ===========================================================================
// 32x32x16-MFMA forward variant (EXPERIMENTAL, measured SLOWER: 570us vs
// 272us at the bench shape).  Wave owns 32 q rows; 16 waves per block.
// Numerically verified (maxdiff 2e-3 vs v1).  Why it loses: 1024-thread
// blocks require 4 waves/SIMD residency, capping VGPR at 128, while the
// f32x16 accumulators want ~150 — the compiler spills 272 B/lane to
// scratch.  The path forward (round 2) is keeping P in registers via
// permlane exchanges to shed the accumulator+LDS round-trip pressure.
// Kept as a measured data point + layout reference (mfma_probe32).
// ===========================================================================
#define ATTN32_THREADS 1024

template <int Dp, bool AL = false>
__global__ __launch_bounds__(ATTN32_THREADS) void attn_fwd_kernel32(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ o,
    float* __restrict__ lse_out, int H, int S, int Skv, int D, float scale,
    int causal, const float* __restrict__ alibi, AttnStrides st) {
  constexpr int KS_QK = Dp / 16;       // k-steps over head dim (K=16)
  constexpr int MT = 2;                // 2 kv m-tiles of 32 per 64-kv tile
  constexpr int DT = Dp / 32;          // d tiles of 32 (O^T rows)
  constexpr int LP = 4;
  constexpr int NW = ATTN32_THREADS / 64;  // 16 waves
  constexpr int GPR = Dp / 8;
  constexpr int TOTAL_G = 64 * GPR;

  const int qb = blockIdx.x;
  const int bh = blockIdx.y;
  const int batch = bh / H, head = bh % H;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lo = lane & 31;            // q column / m row (32-wide)
  const int hi = lane >> 5;            // 0/1

  const short* qp = q + batch * st.qb + head * st.qh;
  const short* kp = k + batch * st.kb + head * st.kh;
  const short* vp = v + batch * st.vb + head * st.vh;
  short* op = o + batch * st.ob + head * st.oh;
  const int q_row0 = qb * (32 * NW) + wave * 32;
  const int my_q = q_row0 + lo;
  const float al_slope = AL ? alibi[head] : 0.f;
  const int al_qoff = Skv - S;

  __shared__ short k_lds[2][64][Dp + LP];
  __shared__ short vt_lds[2][Dp][64 + LP];
  __shared__ short p_lds[NW][32][64 + LP];

  // Q fragments (B-operand): q_frag[ks][j] = Q[my_q][ks*16 + hi*8 + j]
  bf16x8 q_frag[KS_QK];
  {
    int row = min(my_q, S - 1);
#pragma unroll
    for (int ks = 0; ks < KS_QK; ++ks) {
      int col = ks * 16 + hi * 8;
      bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
      q_frag[ks] = z;
      if (col + 8 <= D)
        q_frag[ks] =
            *reinterpret_cast<const bf16x8*>(qp + (int64_t)row * st.qs + col);
    }
  }

  float m_state = -INFINITY, l_state = 0.f;
  f32x16 o_acc[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) o_acc[dt] = f32x16{};

  const int kv_limit = causal ? min(Skv, qb * (32 * NW) + 32 * NW) : Skv;
  const int n_tiles = (kv_limit + 63) / 64;

  bf16x8 kreg[2], vreg[2];
  auto issue_loads = [&](int tile) {
    int t = threadIdx.x;
    bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
    kreg[0] = kreg[1] = vreg[0] = vreg[1] = z;
    if (t < TOTAL_G / 2) {
      int kvr = (t / GPR) * 2, dg = (t % GPR) * 8;
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        int src = tile * 64 + kvr + u;
        if (src < Skv && dg + 8 <= D) {
          kreg[u] = *reinterpret_cast<const bf16x8*>(
              kp + (int64_t)src * st.ks + dg);
          vreg[u] = *reinterpret_cast<const bf16x8*>(
              vp + (int64_t)src * st.vs + dg);
        }
      }
    }
  };
  auto write_tile = [&](int buf) {
    int t = threadIdx.x;
    if (t < TOTAL_G / 2) {
      int kvr = (t / GPR) * 2, dg = (t % GPR) * 8;
      *reinterpret_cast<bf16x8*>(&k_lds[buf][kvr][dg]) = kreg[0];
      *reinterpret_cast<bf16x8*>(&k_lds[buf][kvr + 1][dg]) = kreg[1];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        short2 pr;
        pr.x = vreg[0][j];
        pr.y = vreg[1][j];
        *reinterpret_cast<short2*>(&vt_lds[buf][dg + j][kvr]) = pr;
      }
    }
  };

  if (n_tiles > 0) {
    issue_loads(0);
    write_tile(0);
    __syncthreads();
  }

  for (int ti = 0; ti < n_tiles; ++ti) {
    const int kvb = ti * 64;
    const int cur = ti & 1;
    if (ti + 1 < n_tiles) issue_loads(ti + 1);

    if (causal && kvb > q_row0 + 31) {
      if (ti + 1 < n_tiles) { write_tile(1 - cur); __syncthreads(); }
      continue;
    }

    // ---- S^T = K Q^T: A = K (m=kv), B = q_frag ----
    f32x16 s_acc[MT];
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
      f32x16 acc = {};
#pragma unroll
      for (int ks = 0; ks < KS_QK; ++ks) {
        bf16x8 a = *reinterpret_cast<const bf16x8*>(
            &k_lds[cur][mt * 32 + lo][ks * 16 + hi * 8]);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, q_frag[ks], acc, 0,
                                                      0, 0);
      }
      s_acc[mt] = acc;
    }

    // ---- mask + scale + lane-local max (32 values for q = my_q) ----
    float tile_max = -INFINITY;
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int kv_idx = kvb + mt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float sv = s_acc[mt][r] * scale;
        if (AL) sv = fmaf(al_slope, (float)(kv_idx - my_q - al_qoff), sv);
        bool masked = (kv_idx >= Skv) || (causal && kv_idx > my_q) ||
                      (my_q >= S);
        sv = masked ? -INFINITY : sv;
        s_acc[mt][r] = sv;
        tile_max = fmaxf(tile_max, sv);
      }
    }
    tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32, 64));

    float m_new = fmaxf(m_state, tile_max);
    float alpha = (m_state == -INFINITY) ? 0.f : __expf(m_state - m_new);
    m_state = m_new;
    l_state *= alpha;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r) o_acc[dt][r] *= alpha;

    float part = 0.f;
#pragma unroll
    for (int mt = 0; mt < MT; ++mt) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float pval = (s_acc[mt][r] == -INFINITY)
                         ? 0.f
                         : __expf(s_acc[mt][r] - m_state);
        part += pval;
        p_lds[wave][lo][mt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi] =
            f2bf(pval);
      }
    }
    part += __shfl_xor(part, 32, 64);
    l_state += part;

    // ---- O^T += V^T P^T: A = V^T (m=d), B = P^T (n=q) ----
#pragma unroll
    for (int ks = 0; ks < 64 / 16; ++ks) {
      bf16x8 b = *reinterpret_cast<const bf16x8*>(
          &p_lds[wave][lo][ks * 16 + hi * 8]);
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        bf16x8 a = *reinterpret_cast<const bf16x8*>(
            &vt_lds[cur][dt * 32 + lo][ks * 16 + hi * 8]);
        o_acc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, o_acc[dt],
                                                            0, 0, 0);
      }
    }

    if (ti + 1 < n_tiles) {
      write_tile(1 - cur);
      __syncthreads();
    }
  }

  // ---- epilogue: lane holds O^T[d = dt*32 + (r&3)+8*(r>>2)+4*hi][my_q] --
  if (my_q < S) {
    float inv_l = (l_state > 0.f) ? 1.0f / l_state : 0.f;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int col = dt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        if (col < D)
          op[(int64_t)my_q * st.os + col] = f2bf(o_acc[dt][r] * inv_l);
      }
    }
    if (hi == 0 && lse_out != nullptr)
      lse_out[(int64_t)bh * S + my_q] =
          (l_state > 0.f) ? m_state + __logf(l_state) : -INFINITY;
  }
}

extern "C" {

hipError_t launch_attn_fwd_ablate(const void* q, const void* k,
                                  const void* v, void* o, float* lse,
                                  int64_t B, int64_t H, int64_t S,
                                  int64_t Skv, int64_t D, float scale,
                                  int causal, const int64_t* strides,
                                  int abl, hipStream_t stream) {
  dim3 grid((uint32_t)ceil_div(S, 16 * (ATTN_THREADS / 64)), (uint32_t)(B * H));
  dim3 block(ATTN_THREADS);
  AttnStrides st;
  st.qb = strides[0]; st.qh = strides[1]; st.qs = strides[2];
  st.kb = strides[3]; st.kh = strides[4]; st.ks = strides[5];
  st.vb = strides[6]; st.vh = strides[7]; st.vs = strides[8];
  st.ob = strides[9]; st.oh = strides[10]; st.os = strides[11];
  if (D > 96) return hipErrorInvalidValue;
#define ABL_CASE(A)                                                         \
  if (abl == A) {                                                           \
    attn_fwd_kernel<96, A><<<grid, block, 0, stream>>>(                     \
        (const short*)q, (const short*)k, (const short*)v, (short*)o, lse,  \
        (int)H, (int)S, (int)Skv, (int)D, scale, causal, nullptr, nullptr, \
        st);                                                                \
  }
  ABL_CASE(0) ABL_CASE(1) ABL_CASE(2) ABL_CASE(3) ABL_CASE(4)
#undef ABL_CASE
  return hipGetLastError();
}

hipError_t launch_attn_fwd_v2(const void* q, const void* k, const void* v,
                              void* o, float* lse, int64_t B, int64_t H,
                              int64_t S, int64_t Skv, int64_t D, float scale,
                              int causal, const int64_t* strides,
                              hipStream_t stream) {
  dim3 grid((uint32_t)ceil_div(S, 32 * (ATTN32_THREADS / 64)),
            (uint32_t)(B * H));
  dim3 block(ATTN32_THREADS);
  AttnStrides st;
  st.qb = strides[0]; st.qh = strides[1]; st.qs = strides[2];
  st.kb = strides[3]; st.kh = strides[4]; st.ks = strides[5];
  st.vb = strides[6]; st.vh = strides[7]; st.vs = strides[8];
  st.ob = strides[9]; st.oh = strides[10]; st.os = strides[11];
  if (D <= 64) {
    attn_fwd_kernel32<64><<<grid, block, 0, stream>>>(
        (const short*)q, (const short*)k, (const short*)v, (short*)o, lse,
        (int)H, (int)S, (int)Skv, (int)D, scale, causal, nullptr, st);
  } else if (D <= 96) {
    attn_fwd_kernel32<96><<<grid, block, 0, stream>>>(
        (const short*)q, (const short*)k, (const short*)v, (short*)o, lse,
        (int)H, (int)S, (int)Skv, (int)D, scale, causal, nullptr, st);
  } else if (D <= 128) {
    attn_fwd_kernel32<128><<<grid, block, 0, stream>>>(
        (const short*)q, (const short*)k, (const short*)v, (short*)o, lse,
        (int)H, (int)S, (int)Skv, (int)D, scale, causal, nullptr, st);
  } else {
    return hipErrorInvalidValue;
  }
  return hipGetLastError();
}

static int attn_fwd_nt() {
  // block size selection: 512 threads (8 waves) fits TWO blocks per CU
  // at Dp=96 double-buffered (72.6 KB LDS each) — cross-block overlap
  // hides the per-tile barriers that a single 16-wave block serializes.
  // Override with ALPA_ATTN_FWD_NT=1024 for the round-1 shape.
  static int nt = -1;
  if (nt < 0) {
    const char* e = getenv("ALPA_ATTN_FWD_NT");
    nt = e ? atoi(e) : 512;
    if (nt != 512 && nt != 1024) nt = 512;
  }
  return nt;
}

hipError_t launch_attn_fwd(const void* q, const void* k, const void* v,
                           void* o, float* lse, int64_t B, int64_t H,
                           int64_t S, int64_t Skv, int64_t D, float scale,
                           int causal, const float* alibi,
                           const int* kv_lens, const int64_t* strides,
                           hipStream_t stream) {
  const int NTsel = attn_fwd_nt();
  dim3 grid((uint32_t)ceil_div(S, 16 * (NTsel / 64)), (uint32_t)(B * H));
  dim3 block(NTsel);
#define FWD_VARIANT(DP, ALB, VLB, ALP, VLP)                                \
  do {                                                                     \
    if (NTsel == 512)                                                      \
      attn_fwd_kernel<DP, 0, ALB, VLB, true, 512>                          \
          <<<grid, block, 0, stream>>>(                                    \
              (const short*)q, (const short*)k, (const short*)v,           \
              (short*)o, lse, (int)H, (int)S, (int)Skv, (int)D, scale,     \
              causal, ALP, VLP, st);                                       \
    else                                                                   \
      attn_fwd_kernel<DP, 0, ALB, VLB, true, 1024>                         \
          <<<grid, block, 0, stream>>>(                                    \
              (const short*)q, (const short*)k, (const short*)v,           \
              (short*)o, lse, (int)H, (int)S, (int)Skv, (int)D, scale,     \
              causal, ALP, VLP, st);                                       \
  } while (0)
#define FWD_DISPATCH(DP)                                                   \
  do {                                                                     \
    if (alibi && kv_lens) FWD_VARIANT(DP, true, true, alibi, kv_lens);     \
    else if (alibi) FWD_VARIANT(DP, true, false, alibi, nullptr);          \
    else if (kv_lens) FWD_VARIANT(DP, false, true, nullptr, kv_lens);      \
    else FWD_VARIANT(DP, false, false, nullptr, nullptr);                  \
  } while (0)
  AttnStrides st;
  st.qb = strides[0];
  st.qh = strides[1];
  st.qs = strides[2];
  st.kb = strides[3];
  st.kh = strides[4];
  st.ks = strides[5];
  st.vb = strides[6];
  st.vh = strides[7];
  st.vs = strides[8];
  st.ob = strides[9];
  st.oh = strides[10];
  st.os = strides[11];
  if (D <= 64) {
    FWD_DISPATCH(64);
  } else if (D <= 96) {
    FWD_DISPATCH(96);
  } else if (D <= 128) {
    FWD_DISPATCH(128);
  } else if (D <= 256 && !alibi && !kv_lens) {
    // split-D experiment for CodeGen-class heads: single-buffered K/V
    // (85.5 KB LDS at Dp=256), 512 threads; register tiling carries 16
    // o_acc f32x4 — expect low occupancy, measured against the blocked
    // hipBLASLt path before becoming the default (ROUND2 item 4)
    dim3 grid2((uint32_t)ceil_div(S, 16 * 8), (uint32_t)(B * H));
    attn_fwd_kernel<256, 0, false, false, false, 512>
        <<<grid2, dim3(512), 0, stream>>>(
            (const short*)q, (const short*)k, (const short*)v, (short*)o,
            lse, (int)H, (int)S, (int)Skv, (int)D, scale, causal,
            nullptr, nullptr, st);
  } else {
    return hipErrorInvalidValue;
  }
#undef FWD_DISPATCH
#undef FWD_VARIANT
  return hipGetLastError();
}

// Single-buffer experiment entry (plain path only: no alibi/varlen)
hipError_t launch_attn_fwd_sbuf(const void* q, const void* k, const void* v,
                                void* o, float* lse, int64_t B, int64_t H,
                                int64_t S, int64_t Skv, int64_t D,
                                float scale, int causal,
                                const int64_t* strides,
                                hipStream_t stream) {
  // 512 threads + single-buffered K/V = 43.25 KB LDS at Dp=96 ->
  // THREE blocks per CU (24 waves); cross-block overlap replaces the
  // intra-block prefetch
  dim3 grid((uint32_t)ceil_div(S, 16 * (512 / 64)), (uint32_t)(B * H));
  dim3 block(512);
  AttnStrides st;
  st.qb = strides[0]; st.qh = strides[1]; st.qs = strides[2];
  st.kb = strides[3]; st.kh = strides[4]; st.ks = strides[5];
  st.vb = strides[6]; st.vh = strides[7]; st.vs = strides[8];
  st.ob = strides[9]; st.oh = strides[10]; st.os = strides[11];
#define SBUF_CASE(DP)                                                      \
  attn_fwd_kernel<DP, 0, false, false, false, 512>                         \
      <<<grid, block, 0, stream>>>(                                        \
      (const short*)q, (const short*)k, (const short*)v, (short*)o, lse,   \
      (int)H, (int)S, (int)Skv, (int)D, scale, causal, nullptr, nullptr,   \
      st)
  if (D <= 64) SBUF_CASE(64);
  else if (D <= 96) SBUF_CASE(96);
  else if (D <= 128) SBUF_CASE(128);
  else return hipErrorInvalidValue;
#undef SBUF_CASE
  return hipGetLastError();
}

}  // extern "C"

// ---------------------------------------------------------------------------
// MFMA layout probe: D = A @ B for a single 16x32 x 32x16 tile, to verify
// the fragment layouts above against a CPU reference on real hardware.
__global__ void mfma_probe_kernel(const short* __restrict__ a,
                                  const short* __restrict__ b,
                                  float* __restrict__ d) {
  int lane = threadIdx.x & 63;
  int lo = lane & 15, hi = lane >> 4;
  bf16x8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = a[lo * 32 + hi * 8 + j];       // A[m][k] row-major 16x32
    bf[j] = b[(hi * 8 + j) * 16 + lo];     // B[k][n] row-major 32x16
  }
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) d[(hi * 4 + r) * 16 + lo] = acc[r];
}

extern "C" hipError_t launch_mfma_probe(const void* a, const void* b,
                                        float* d, hipStream_t stream) {
  mfma_probe_kernel<<<dim3(1), dim3(64), 0, stream>>>((const short*)a,
                                                      (const short*)b, d);
  return hipGetLastError();
}

// 32x32x16 layout probe: D(32x32) = A(32x16) @ B(16x32).
// Assumed layouts: A: m=lane&31, k=(lane>>5)*8+j; B: n=lane&31, same k;
// C/D (documented, guide §3): col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5)
__global__ void mfma_probe32_kernel(const short* __restrict__ a,
                                    const short* __restrict__ b,
                                    float* __restrict__ d) {
  int lane = threadIdx.x & 63;
  int lo = lane & 31, hi = lane >> 5;
  bf16x8 af, bf;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    af[j] = a[lo * 16 + hi * 8 + j];
    bf[j] = b[(hi * 8 + j) * 32 + lo];
  }
  f32x16 acc = {};
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(af, bf, acc, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int row = (r & 3) + 8 * (r >> 2) + 4 * hi;
    d[row * 32 + lo] = acc[r];
  }
}

extern "C" hipError_t launch_mfma_probe32(const void* a, const void* b,
                                          float* d, hipStream_t stream) {
  mfma_probe32_kernel<<<dim3(1), dim3(64), 0, stream>>>((const short*)a,
                                                        (const short*)b, d);
  return hipGetLastError();
}

// ===========================================================================
// Flash attention BACKWARD (bf16, causal/full), deterministic two-kernel
// split (no atomics): dq kernel over q-blocks, dk/dv kernel over kv-blocks,
// each recomputing P from (q, k, lse).  Same MFMA fragment layouts as fwd.
//
// delta = rowsum(dO * O) is precomputed by attn_bwd_preprocess.
// ===========================================================================

// All tensors logically [B, H, S, D] with arbitrary B/H/S strides and a
// contiguous D (so the packed qkv/dqkv layouts feed the kernels directly).
struct AttnBwdStrides {
  int64_t qb, qh, qs;
  int64_t kb, kh, ks;
  int64_t vb, vh, vs;
  int64_t dob, doh, dos;
  int64_t dqb, dqh, dqs;
  int64_t dkb, dkh, dks;
  int64_t dvb, dvh, dvs;
};

__global__ void attn_bwd_preprocess_kernel(const short* __restrict__ dout,
                                           const short* __restrict__ o,
                                           float* __restrict__ delta,
                                           int64_t rows, int H, int S, int D,
                                           int64_t dob, int64_t doh,
                                           int64_t dos, int64_t ob,
                                           int64_t oh, int64_t os) {
  // 16 lanes x bf16x8 per row, 4 rows per wave (the one-wave-per-row
  // version left 44/64 lanes idle at D=80 and ran 6.7x off roofline)
  int64_t row = (int64_t)blockIdx.x * (blockDim.x >> 4) + (threadIdx.x >> 4);
  if (row >= rows) return;
  int lane16 = threadIdx.x & 15;
  int64_t sidx = row % S, h = (row / S) % H, b = row / ((int64_t)S * H);
  const short* dr = dout + b * dob + h * doh + sidx * dos;
  const short* orow = o + b * ob + h * oh + sidx * os;
  float s = 0.f;
  for (int i = lane16 * 8; i + 8 <= D; i += 16 * 8) {
    bf16x8 dv = *reinterpret_cast<const bf16x8*>(dr + i);
    bf16x8 ov = *reinterpret_cast<const bf16x8*>(orow + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) s += bf2f(dv[j]) * bf2f(ov[j]);
  }
  // D not a multiple of 8: scalar tail on lane 0
  if (lane16 == 0)
    for (int i = (D / 8) * 8; i < D; ++i)
      s += bf2f(dr[i]) * bf2f(orow[i]);
#pragma unroll
  for (int off = 8; off > 0; off >>= 1) s += __shfl_xor(s, off, 64);
  if (lane16 == 0) delta[row] = s;
}

// ---------------------------------------------------------------- dq kernel
// Block: 4 waves, 64 q rows; loops kv blocks of 64.
//   S = Q K^T          (A=Q regs, B=K_lds row-major)
//   P = exp(S*? - lse) (lse already includes the scale from fwd)
//   dP = dO V^T        (A=dO regs, B=V_lds row-major)
//   dS = P*(dP-delta)*scale
//   dQ += dS K         (A=dS via p_lds, B=Kt_lds transposed)
template <int Dp, bool AL = false>
__global__ __launch_bounds__(ATTN_BWD_THREADS) void attn_bwd_dq_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dq, int H, int S, int Skv, int D, float scale,
    int causal, const float* __restrict__ alibi, AttnBwdStrides st) {
  // 8 waves x 16 q rows = 128 q rows per block; K/V/K^T staged per kv tile
  // with register-prefetched loads (issue before compute, write after the
  // barrier) and paired-row transposes.
  constexpr int KD = Dp / 32;
  constexpr int NT = 4;                 // 64 kv per tile
  constexpr int DT = Dp / 16;
  constexpr int LP = 4;
  constexpr int NW = ATTN_BWD_THREADS / 64;
  constexpr int GPR = Dp / 8;
  constexpr int PAIRS = 32 * GPR;       // (64 rows / 2) * groups

  const int flat_ = xcd_swizzle(
      (int)(blockIdx.y * gridDim.x + blockIdx.x),
      (int)(gridDim.x * gridDim.y));
  const int qb_raw = flat_ % (int)gridDim.x;
  const int qb = causal ? ((int)gridDim.x - 1 - qb_raw) : qb_raw;
  const int bh = flat_ / (int)gridDim.x;
  const int batch = bh / H, head = bh % H;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lo = lane & 15, hi = lane >> 4;

  const short* qp = q + batch * st.qb + head * st.qh;
  const short* kp = k + batch * st.kb + head * st.kh;
  const short* vp = v + batch * st.vb + head * st.vh;
  const short* dop = dout + batch * st.dob + head * st.doh;
  short* dqp = dq + batch * st.dqb + head * st.dqh;
  const int q_row0 = qb * (16 * NW) + wave * 16;

  __shared__ short k_lds[64][Dp + LP];
  __shared__ short kt_lds[Dp][64 + LP];
  __shared__ short v_lds[64][Dp + LP];
  __shared__ short p_lds[NW][16][64 + LP];

  bf16x8 q_frag[KD], do_frag[KD];
  {
    int row = min(q_row0 + lo, S - 1);
#pragma unroll
    for (int ks = 0; ks < KD; ++ks) {
      int col = ks * 32 + hi * 8;
      bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
      q_frag[ks] = z;
      do_frag[ks] = z;
      if (col + 8 <= D) {
        q_frag[ks] = *reinterpret_cast<const bf16x8*>(
            qp + (int64_t)row * st.qs + col);
        do_frag[ks] = *reinterpret_cast<const bf16x8*>(
            dop + (int64_t)row * st.dos + col);
      }
    }
  }
  float lse_r[4], delta_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int qi = q_row0 + hi * 4 + r;
    lse_r[r] = (qi < S) ? lse[(int64_t)bh * S + qi] : 0.f;
    delta_r[r] = (qi < S) ? delta[(int64_t)bh * S + qi] : 0.f;
  }
  const float al_slope = AL ? alibi[bh % H] : 0.f;

  f32x4 dq_acc[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) dq_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kv_limit = causal ? min(Skv, qb * (16 * NW) + 16 * NW) : Skv;
  const int n_tiles = (kv_limit + 63) / 64;

  bf16x8 kreg[2], vreg[2];
  auto issue_loads = [&](int tile) {
    int t = threadIdx.x;
    bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
    kreg[0] = kreg[1] = vreg[0] = vreg[1] = z;
    if (t < PAIRS) {
      int kvr = (t / GPR) * 2, dg = (t % GPR) * 8;
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        int src = tile * 64 + kvr + u;
        if (src < Skv && dg + 8 <= D) {
          kreg[u] = *reinterpret_cast<const bf16x8*>(
              kp + (int64_t)src * st.ks + dg);
          vreg[u] = *reinterpret_cast<const bf16x8*>(
              vp + (int64_t)src * st.vs + dg);
        }
      }
    }
  };
  auto write_tile = [&]() {
    int t = threadIdx.x;
    if (t < PAIRS) {
      int kvr = (t / GPR) * 2, dg = (t % GPR) * 8;
      *reinterpret_cast<bf16x8*>(&k_lds[kvr][dg]) = kreg[0];
      *reinterpret_cast<bf16x8*>(&k_lds[kvr + 1][dg]) = kreg[1];
      *reinterpret_cast<bf16x8*>(&v_lds[kvr][dg]) = vreg[0];
      *reinterpret_cast<bf16x8*>(&v_lds[kvr + 1][dg]) = vreg[1];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        short2 pr;
        pr.x = kreg[0][j];
        pr.y = kreg[1][j];
        *reinterpret_cast<short2*>(&kt_lds[dg + j][kvr]) = pr;
      }
    }
  };

  if (n_tiles > 0) {
    issue_loads(0);
    write_tile();
    __syncthreads();
  }

  for (int ti = 0; ti < n_tiles; ++ti) {
    const int kvb = ti * 64;
    if (ti + 1 < n_tiles) issue_loads(ti + 1);

    // causal: kv tiles entirely above this wave's q rows are masked out
    if (causal && kvb > q_row0 + 15) {
      __syncthreads();
      if (ti + 1 < n_tiles) { write_tile(); __syncthreads(); }
      continue;
    }

    // S and dP tiles (C: row = q (hi*4+r), col = kv (nt*16+lo))
__builtin_amdgcn_s_setprio(1);  // T5: favor the MFMA cluster
    f32x4 s_acc[NT], dp_acc[NT];
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
      f32x4 sa = {0.f, 0.f, 0.f, 0.f}, da = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < KD; ++ks) {
        bf16x8 bk = *reinterpret_cast<const bf16x8*>(
            &k_lds[nt * 16 + lo][ks * 32 + hi * 8]);
        bf16x8 bv = *reinterpret_cast<const bf16x8*>(
            &v_lds[nt * 16 + lo][ks * 32 + hi * 8]);
        sa = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[ks], bk, sa, 0,
                                                     0, 0);
        da = __builtin_amdgcn_mfma_f32_16x16x32_bf16(do_frag[ks], bv, da, 0,
                                                     0, 0);
      }
      s_acc[nt] = sa;
      dp_acc[nt] = da;
    }
    __builtin_amdgcn_s_setprio(0);

    // dS = P * (dP - delta) * scale -> p_lds as the dS A-operand
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
      int kv_idx = kvb + nt * 16 + lo;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int q_idx = q_row0 + hi * 4 + r;
        bool masked = (kv_idx >= Skv) || (causal && kv_idx > q_idx) ||
                      (q_idx >= S);
        float sraw = s_acc[nt][r] * scale;
        if (AL) sraw = fmaf(al_slope, (float)(kv_idx - q_idx), sraw);
        float pv = masked ? 0.f : __expf(sraw - lse_r[r]);
        float ds = pv * (dp_acc[nt][r] - delta_r[r]) * scale;
        p_lds[wave][hi * 4 + r][nt * 16 + lo] = f2bf(ds);
      }
    }

__builtin_amdgcn_s_setprio(1);  // T5: favor the MFMA cluster
    // dQ += dS @ K  (A = dS from p_lds, B = Kt)
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &p_lds[wave][lo][ks * 32 + hi * 8]);
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &kt_lds[dt * 16 + lo][ks * 32 + hi * 8]);
        dq_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, b, dq_acc[dt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    __syncthreads();
    if (ti + 1 < n_tiles) {
      write_tile();
      __syncthreads();
    }
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int q_idx = q_row0 + hi * 4 + r;
    if (q_idx >= S) continue;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      int col = dt * 16 + lo;
      if (col < D)
        dqp[(int64_t)q_idx * st.dqs + col] = f2bf(dq_acc[dt][r]);
    }
  }
}

// ------------------------------------------------------------- dk/dv kernel
// Block: 4 waves, 64 kv rows; loops q blocks of 64 (from the diagonal for
// causal).
//   S^T = K Q^T        (A=K regs, B=Q_lds row-major)
//   P^T = exp(S^T*scale - lse[q])
//   dV += P^T dO       (A=P^T via p_lds, B=dOt_lds)
//   dP^T = V dO^T      (A=V regs, B=dO_lds row-major)
//   dS^T = P^T*(dP^T - delta[q])*scale
//   dK += dS^T Q       (A=dS^T via p_lds, B=Qt_lds)
template <int Dp, int ABL = 0, bool AL = false, bool LEAN = false>
__global__ __launch_bounds__(ATTN_BWD_THREADS) void attn_bwd_dkv_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dk, short* __restrict__ dv, int H, int S, int Skv,
    int D, float scale, int causal, const float* __restrict__ alibi,
    AttnBwdStrides st) {
  // 8 waves x 16 kv rows = 128 kv rows per block; Q/dO tiles (row-major +
  // transposed) staged per q tile with register-prefetched loads and
  // paired-row transposes — the staging cost is the dominant term here and
  // is amortized over 2x the kv rows vs the 4-wave version.
  constexpr int KD = Dp / 32;
  constexpr int NT = 4;   // 64 q per tile
  constexpr int DT = Dp / 16;
  constexpr int LP = 4;
  constexpr int NW = ATTN_BWD_THREADS / 64;
  constexpr int GPR = Dp / 8;
  constexpr int PAIRS = 32 * GPR;

  const int flat_ = xcd_swizzle(
      (int)(blockIdx.y * gridDim.x + blockIdx.x),
      (int)(gridDim.x * gridDim.y));
  const int kvb_idx = flat_ % (int)gridDim.x;
  const int bh = flat_ / (int)gridDim.x;
  const int batch = bh / H, head = bh % H;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lo = lane & 15, hi = lane >> 4;

  const short* qp = q + batch * st.qb + head * st.qh;
  const short* kp = k + batch * st.kb + head * st.kh;
  const short* vp = v + batch * st.vb + head * st.vh;
  const short* dop = dout + batch * st.dob + head * st.doh;
  short* dkp = dk + batch * st.dkb + head * st.dkh;
  short* dvp = dv + batch * st.dvb + head * st.dvh;
  const int kv_row0 = kvb_idx * (16 * NW) + wave * 16;
  const float al_slope = AL ? alibi[bh % H] : 0.f;

  __shared__ short q_lds[64][Dp + LP];
  __shared__ short do_lds[64][Dp + LP];
  // LEAN: drop the transposed Q/dO tiles (-27.6 KB LDS => 3 blocks/CU)
  // and gather the dV/dK B-fragments from the row-major tiles with
  // strided b16 reads instead (occupancy-vs-LDS-op-count experiment)
  __shared__ short qt_lds[LEAN ? 1 : Dp][64 + LP];
  __shared__ short dot_lds[LEAN ? 1 : Dp][64 + LP];
  __shared__ short p_lds[NW][16][64 + LP];
  __shared__ float lse_lds[64];
  __shared__ float delta_lds[64];

  bf16x8 k_frag[KD], v_frag[KD];
  {
    int row = min(kv_row0 + lo, Skv - 1);
#pragma unroll
    for (int ks = 0; ks < KD; ++ks) {
      int col = ks * 32 + hi * 8;
      bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
      k_frag[ks] = z;
      v_frag[ks] = z;
      if (col + 8 <= D) {
        k_frag[ks] = *reinterpret_cast<const bf16x8*>(
            kp + (int64_t)row * st.ks + col);
        v_frag[ks] = *reinterpret_cast<const bf16x8*>(
            vp + (int64_t)row * st.vs + col);
      }
    }
  }

  f32x4 dk_acc[DT], dv_acc[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) {
    dk_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};
    dv_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};
  }

  const int q_start = causal ? kvb_idx * (16 * NW) / 64 * 64 : 0;
  const int n_tiles = (S - q_start + 63) / 64;

  bf16x8 qreg[2], doreg[2];
  float lse_reg, delta_reg;
  auto issue_loads = [&](int tile) {
    int t = threadIdx.x;
    bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
    qreg[0] = qreg[1] = doreg[0] = doreg[1] = z;
    int qb0 = q_start + tile * 64;
    if (t < PAIRS) {
      int qr = (t / GPR) * 2, dg = (t % GPR) * 8;
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        int src = qb0 + qr + u;
        if (src < S && dg + 8 <= D) {
          qreg[u] = *reinterpret_cast<const bf16x8*>(
              qp + (int64_t)src * st.qs + dg);
          doreg[u] = *reinterpret_cast<const bf16x8*>(
              dop + (int64_t)src * st.dos + dg);
        }
      }
    }
    lse_reg = delta_reg = 0.f;
    if (t < 64) {
      int src = qb0 + t;
      if (src < S) {
        lse_reg = lse[(int64_t)bh * S + src];
        delta_reg = delta[(int64_t)bh * S + src];
      }
    }
  };
  auto write_tile = [&]() {
    int t = threadIdx.x;
    if (t < PAIRS) {
      int qr = (t / GPR) * 2, dg = (t % GPR) * 8;
      *reinterpret_cast<bf16x8*>(&q_lds[qr][dg]) = qreg[0];
      *reinterpret_cast<bf16x8*>(&q_lds[qr + 1][dg]) = qreg[1];
      *reinterpret_cast<bf16x8*>(&do_lds[qr][dg]) = doreg[0];
      *reinterpret_cast<bf16x8*>(&do_lds[qr + 1][dg]) = doreg[1];
      if (!LEAN) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          short2 pq, pd;
          pq.x = qreg[0][j];
          pq.y = qreg[1][j];
          pd.x = doreg[0][j];
          pd.y = doreg[1][j];
          *reinterpret_cast<short2*>(&qt_lds[dg + j][qr]) = pq;
          *reinterpret_cast<short2*>(&dot_lds[dg + j][qr]) = pd;
        }
      }
    }
    if (t < 64) {
      lse_lds[t] = lse_reg;
      delta_lds[t] = delta_reg;
    }
  };

  if (n_tiles > 0) {
    issue_loads(0);
    write_tile();
    __syncthreads();
  }

  for (int ti = 0; ti < n_tiles; ++ti) {
    const int qb0 = q_start + ti * 64;
    if (ti + 1 < n_tiles) issue_loads(ti + 1);

    // causal: q tiles entirely before this wave's kv rows are masked out
    if (causal && qb0 + 63 < kv_row0) {
      __syncthreads();
      if (ti + 1 < n_tiles) { write_tile(); __syncthreads(); }
      continue;
    }

    if (ABL >= 3) {  // staging only
      float keep = bf2f(q_lds[lane & 63][0]) + bf2f(do_lds[0][lane & 63]) +
                   bf2f(q_lds[0][lane & 63]) + bf2f(do_lds[lane & 63][0]);
      asm volatile("" ::"v"(keep));
      __syncthreads();
      if (ti + 1 < n_tiles) { write_tile(); __syncthreads(); }
      continue;
    }
__builtin_amdgcn_s_setprio(1);  // T5
    // S^T and dP^T tiles (C: row = kv (hi*4+r), col = q (nt*16+lo))
    f32x4 st_acc[NT], dpt_acc[NT];
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
      f32x4 sa = {0.f, 0.f, 0.f, 0.f}, da = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < KD; ++ks) {
        bf16x8 bq = *reinterpret_cast<const bf16x8*>(
            &q_lds[nt * 16 + lo][ks * 32 + hi * 8]);
        bf16x8 bd = *reinterpret_cast<const bf16x8*>(
            &do_lds[nt * 16 + lo][ks * 32 + hi * 8]);
        sa = __builtin_amdgcn_mfma_f32_16x16x32_bf16(k_frag[ks], bq, sa, 0,
                                                     0, 0);
        da = __builtin_amdgcn_mfma_f32_16x16x32_bf16(v_frag[ks], bd, da, 0,
                                                     0, 0);
      }
      st_acc[nt] = sa;
      dpt_acc[nt] = da;
    }
    __builtin_amdgcn_s_setprio(0);

    // P^T written to p_lds immediately; dS^T retained PACKED as bf16
    // (8 VGPRs, not 16 floats) — keeps the kernel under the 128-VGPR
    // occupancy cliff (166 VGPR cost 1 wave/SIMD before this)
    short dst_pk[NT][4];
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
      int q_idx = qb0 + nt * 16 + lo;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int kv_idx = kv_row0 + hi * 4 + r;
        bool masked = (q_idx >= S) || (kv_idx >= Skv) ||
                      (causal && kv_idx > q_idx);
        float sraw = st_acc[nt][r] * scale;
        if (AL) sraw = fmaf(al_slope, (float)(kv_idx - q_idx), sraw);
        float pv = masked ? 0.f : __expf(sraw - lse_lds[nt * 16 + lo]);
        p_lds[wave][hi * 4 + r][nt * 16 + lo] = f2bf(pv);
        dst_pk[nt][r] =
            f2bf(pv * (dpt_acc[nt][r] - delta_lds[nt * 16 + lo]) * scale);
      }
    }

    if (ABL >= 2) {  // S^T/dP^T + softmax math only
#pragma unroll
      for (int nt = 0; nt < NT; ++nt)
        asm volatile("" ::"v"((int)dst_pk[nt][0]), "v"((int)dst_pk[nt][3]));
      __syncthreads();
      if (ti + 1 < n_tiles) { write_tile(); __syncthreads(); }
      continue;
    }
__builtin_amdgcn_s_setprio(1);  // T5
    // dV += P^T @ dO (A = P^T via p_lds, B = dOt).  The dS^T spill for
    // the dK group is INTERLEAVED into this loop: after dV's ks-th read
    // drains q-columns [32ks, 32ks+32), those p_lds columns are dead and
    // dS^T for tiles nt = 2ks, 2ks+1 overwrites them (same-wave LDS ops
    // complete in issue order, so no barrier is needed).  Measured
    // NEUTRAL on the bench shape — the ablation's "+206 us dK phase" is
    // tail LATENCY (any last MFMA group absorbs the drain into the
    // barrier), not a write->read stall — kept because it frees the
    // dst_pk registers before the dK group.  The remaining bwd levers
    // are structural: 3 blocks/CU via T10 hardware-transpose reads of
    // q/do (dropping qt/dot tiles, -27.6 KB LDS), or FA2-style dq
    // accumulation by atomics inside this kernel (kills the separate dq
    // kernel's S/dP recompute at the cost of fp32 atomics).
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &p_lds[wave][lo][ks * 32 + hi * 8]);
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        bf16x8 b;
        if (LEAN) {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            b[j] = do_lds[ks * 32 + hi * 8 + j][dt * 16 + lo];
        } else {
          b = *reinterpret_cast<const bf16x8*>(
              &dot_lds[dt * 16 + lo][ks * 32 + hi * 8]);
        }
        dv_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, b, dv_acc[dt], 0, 0, 0);
      }
      if (ABL == 0) {
#pragma unroll
        for (int nt = 2 * ks; nt < 2 * ks + 2; ++nt)
#pragma unroll
          for (int r = 0; r < 4; ++r)
            p_lds[wave][hi * 4 + r][nt * 16 + lo] = dst_pk[nt][r];
      }
    }
    __builtin_amdgcn_s_setprio(0);

    if (ABL >= 1) {  // skip the dK group
      __syncthreads();
      if (ti + 1 < n_tiles) { write_tile(); __syncthreads(); }
      continue;
    }
    // dK += dS^T @ Q (A = dS^T via p_lds, B = Qt)
__builtin_amdgcn_s_setprio(1);  // T5
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &p_lds[wave][lo][ks * 32 + hi * 8]);
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        bf16x8 b;
        if (LEAN) {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            b[j] = q_lds[ks * 32 + hi * 8 + j][dt * 16 + lo];
        } else {
          b = *reinterpret_cast<const bf16x8*>(
              &qt_lds[dt * 16 + lo][ks * 32 + hi * 8]);
        }
        dk_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, b, dk_acc[dt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    __syncthreads();
    if (ti + 1 < n_tiles) {
      write_tile();
      __syncthreads();
    }
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int kv_idx = kv_row0 + hi * 4 + r;
    if (kv_idx >= Skv) continue;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      int col = dt * 16 + lo;
      if (col < D) {
        dkp[(int64_t)kv_idx * st.dks + col] = f2bf(dk_acc[dt][r]);
        dvp[(int64_t)kv_idx * st.dvs + col] = f2bf(dv_acc[dt][r]);
      }
    }
  }
}

extern "C" {

hipError_t launch_attn_bwd_dkv_ablate(const void* q, const void* k,
                                      const void* v, const void* dout,
                                      const float* lse, const float* delta,
                                      void* dk, void* dv, int64_t B,
                                      int64_t H, int64_t S, int64_t Skv,
                                      int64_t D, float scale, int causal,
                                      const int64_t* strides, int abl,
                                      hipStream_t stream) {
  AttnBwdStrides st;
  const int64_t* pp = strides;
  st.qb = pp[0]; st.qh = pp[1]; st.qs = pp[2];
  st.kb = pp[3]; st.kh = pp[4]; st.ks = pp[5];
  st.vb = pp[6]; st.vh = pp[7]; st.vs = pp[8];
  st.dob = pp[9]; st.doh = pp[10]; st.dos = pp[11];
  st.dqb = pp[12]; st.dqh = pp[13]; st.dqs = pp[14];
  st.dkb = pp[15]; st.dkh = pp[16]; st.dks = pp[17];
  st.dvb = pp[18]; st.dvh = pp[19]; st.dvs = pp[20];
  dim3 block(ATTN_BWD_THREADS);
  dim3 grid_kv((uint32_t)ceil_div(Skv, 16 * (ATTN_BWD_THREADS / 64)),
               (uint32_t)(B * H));
  if (D > 96) return hipErrorInvalidValue;
#define DKV_CASE(A)                                                        \
  if (abl == A)                                                            \
    attn_bwd_dkv_kernel<96, A><<<grid_kv, block, 0, stream>>>(             \
        (const short*)q, (const short*)k, (const short*)v,                 \
        (const short*)dout, lse, delta, (short*)dk, (short*)dv, (int)H,    \
        (int)S, (int)Skv, (int)D, scale, causal, nullptr, st);
  DKV_CASE(0) DKV_CASE(1) DKV_CASE(2) DKV_CASE(3)
#undef DKV_CASE
  return hipGetLastError();
}

hipError_t launch_attn_bwd(const void* q, const void* k, const void* v,
                           const void* o, const void* dout,
                           const float* lse, float* delta_ws, void* dq,
                           void* dk, void* dv, int64_t B, int64_t H,
                           int64_t S, int64_t Skv, int64_t D, float scale,
                           int causal, const float* alibi,
                           const int64_t* strides, hipStream_t stream) {
  // strides layout: q(3) k(3) v(3) do(3) dq(3) dk(3) dv(3) o(3)
  AttnBwdStrides st;
  const int64_t* p = strides;
  st.qb = p[0]; st.qh = p[1]; st.qs = p[2];
  st.kb = p[3]; st.kh = p[4]; st.ks = p[5];
  st.vb = p[6]; st.vh = p[7]; st.vs = p[8];
  st.dob = p[9]; st.doh = p[10]; st.dos = p[11];
  st.dqb = p[12]; st.dqh = p[13]; st.dqs = p[14];
  st.dkb = p[15]; st.dkh = p[16]; st.dks = p[17];
  st.dvb = p[18]; st.dvh = p[19]; st.dvs = p[20];
  int64_t rows = B * H * S;
  {
    dim3 block(256);
    dim3 grid((uint32_t)ceil_div(rows, 16));  // 16 rows per 256-thr block
    attn_bwd_preprocess_kernel<<<grid, block, 0, stream>>>(
        (const short*)dout, (const short*)o, delta_ws, rows, (int)H, (int)S,
        (int)D, st.dob, st.doh, st.dos, p[21], p[22], p[23]);
  }
  // dq and dkv are independent (disjoint outputs, shared read-only
  // inputs) and each runs at ~3 waves/SIMD — launching them on two
  // streams lets the CU scheduler co-resident them and fill the SIMDs.
  static hipStream_t side_stream = nullptr;
  static hipEvent_t ev_fork = nullptr, ev_join = nullptr;
  if (side_stream == nullptr) {
    hipStreamCreateWithFlags(&side_stream, hipStreamNonBlocking);
    hipEventCreateWithFlags(&ev_fork, hipEventDisableTiming);
    hipEventCreateWithFlags(&ev_join, hipEventDisableTiming);
  }
  hipEventRecord(ev_fork, stream);
  hipStreamWaitEvent(side_stream, ev_fork, 0);
  dim3 block(ATTN_BWD_THREADS);
  dim3 grid_q((uint32_t)ceil_div(S, 16 * (ATTN_BWD_THREADS / 64)),
              (uint32_t)(B * H));
  dim3 grid_kv((uint32_t)ceil_div(Skv, 16 * (ATTN_BWD_THREADS / 64)),
               (uint32_t)(B * H));
#define LAUNCH_BWD_T(DP, ALB, ALP)                                         \
  do {                                                                       \
    attn_bwd_dq_kernel<DP, ALB><<<grid_q, block, 0, side_stream>>>(          \
        (const short*)q, (const short*)k, (const short*)v,                   \
        (const short*)dout, lse, delta_ws, (short*)dq, (int)H, (int)S,       \
        (int)Skv, (int)D, scale, causal, ALP, st);                           \
    if (getenv("ALPA_ATTN_DKV_LEAN") &&                                      \
        getenv("ALPA_ATTN_DKV_LEAN")[0] == '1')                              \
      attn_bwd_dkv_kernel<DP, 0, ALB, true>                                  \
          <<<grid_kv, block, 0, stream>>>(                                   \
              (const short*)q, (const short*)k, (const short*)v,             \
              (const short*)dout, lse, delta_ws, (short*)dk, (short*)dv,     \
              (int)H, (int)S, (int)Skv, (int)D, scale, causal, ALP, st);     \
    else                                                                     \
      attn_bwd_dkv_kernel<DP, 0, ALB><<<grid_kv, block, 0, stream>>>(        \
        (const short*)q, (const short*)k, (const short*)v,                   \
        (const short*)dout, lse, delta_ws, (short*)dk, (short*)dv, (int)H,   \
        (int)S, (int)Skv, (int)D, scale, causal, ALP, st);                   \
  } while (0)
#define LAUNCH_BWD(DP)                                                       \
  do {                                                                       \
    if (alibi) LAUNCH_BWD_T(DP, true, alibi);                                \
    else LAUNCH_BWD_T(DP, false, nullptr);                                   \
  } while (0)
  if (D <= 64) {
    LAUNCH_BWD(64);
  } else if (D <= 96) {
    LAUNCH_BWD(96);
  } else if (D <= 128) {
    LAUNCH_BWD(128);
  } else {
    return hipErrorInvalidValue;
  }
#undef LAUNCH_BWD
  hipEventRecord(ev_join, side_stream);
  hipStreamWaitEvent(stream, ev_join, 0);
  return hipGetLastError();
}

}  // extern "C"
