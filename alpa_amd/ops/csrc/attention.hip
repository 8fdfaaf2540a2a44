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

#define ATTN_BLOCK_Q 64
#define ATTN_BLOCK_K 64
#define ATTN_THREADS 256

// Strides are in elements; the innermost (D) dim must be contiguous.
// Strided addressing lets the packed qkv layout [B, S, heads, 3*D] feed the
// kernel directly — no permute/contiguous copies on the hot path.
struct AttnStrides {
  int64_t qb, qh, qs;  // q batch/head/seq strides
  int64_t kb, kh, ks;
  int64_t vb, vh, vs;
  int64_t ob, oh, os;
};

template <int Dp>
__global__ __launch_bounds__(ATTN_THREADS) void attn_fwd_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ o,
    float* __restrict__ lse_out, int H, int S, int Skv, int D, float scale,
    int causal, AttnStrides st) {
  constexpr int KSTEPS_QK = Dp / 32;   // k-steps over head dim
  constexpr int NTILES = ATTN_BLOCK_K / 16;  // 4
  constexpr int DTILES = Dp / 16;

  const int qb = blockIdx.x;           // q block index
  const int bh = blockIdx.y;           // fused batch*head
  const int batch = bh / H, head = bh % H;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lo = lane & 15;            // fragment row/col low index
  const int hi = lane >> 4;            // fragment quartet index

  const short* qp = q + batch * st.qb + head * st.qh;
  const short* kp = k + batch * st.kb + head * st.kh;
  const short* vp = v + batch * st.vb + head * st.vh;
  short* op = o + batch * st.ob + head * st.oh;
  const int q_row0 = qb * ATTN_BLOCK_Q + wave * 16;  // wave's first q row

  // +8 bf16 (16 B) row padding: unpadded strides are multiples of 128 B,
  // putting all 16 fragment-read lanes in the same LDS bank (8-16-way
  // conflict); the pad keeps 16 B alignment while spreading banks
  // (guide §6 Guideline 4).
  constexpr int LP = 8;
  __shared__ short k_lds[ATTN_BLOCK_K][Dp + LP];
  __shared__ short vt_lds[Dp][ATTN_BLOCK_K + LP];
  __shared__ short p_lds[4][16][ATTN_BLOCK_K + LP];  // per-wave P tile

  // ---- load Q fragments (held in registers for the whole kv loop) ----
  bf16x8 q_frag[KSTEPS_QK];
  {
    int m = q_row0 + lo;
    int row = min(m, S - 1);
#pragma unroll
    for (int ks = 0; ks < KSTEPS_QK; ++ks) {
      int col = ks * 32 + hi * 8;
      if (col + 8 <= D) {
        q_frag[ks] =
            *reinterpret_cast<const bf16x8*>(qp + (int64_t)row * st.qs + col);
      } else {
        bf16x8 z = {0, 0, 0, 0, 0, 0, 0, 0};
        q_frag[ks] = z;
      }
    }
  }

  // ---- online softmax state (per owned row r = hi*4 + reg... here the
  // wave's 16 rows map: reg r of C holds row hi*4+r; every lane tracks the
  // 4 rows of its quartet) ----
  float m_state[4], l_state[4];
  f32x4 o_acc[DTILES];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_state[r] = -INFINITY;
    l_state[r] = 0.f;
  }
#pragma unroll
  for (int dt = 0; dt < DTILES; ++dt) o_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kv_limit =
      causal ? min(Skv, qb * ATTN_BLOCK_Q + ATTN_BLOCK_Q) : Skv;

  for (int kvb = 0; kvb < kv_limit; kvb += ATTN_BLOCK_K) {
    // ---- stage K tile and transposed V tile (all 256 threads) ----
    __syncthreads();
    {
      constexpr int GROUPS_PER_ROW = Dp / 8;
      constexpr int TOTAL = ATTN_BLOCK_K * GROUPS_PER_ROW;
      for (int t = threadIdx.x; t < TOTAL; t += ATTN_THREADS) {
        int kvr = t / GROUPS_PER_ROW;
        int dg = (t % GROUPS_PER_ROW) * 8;
        bf16x8 kv8 = {0, 0, 0, 0, 0, 0, 0, 0};
        bf16x8 vv8 = {0, 0, 0, 0, 0, 0, 0, 0};
        int src = kvb + kvr;
        if (src < Skv && dg + 8 <= D) {
          kv8 = *reinterpret_cast<const bf16x8*>(kp + (int64_t)src * st.ks + dg);
          vv8 = *reinterpret_cast<const bf16x8*>(vp + (int64_t)src * st.vs + dg);
        }
        *reinterpret_cast<bf16x8*>(&k_lds[kvr][dg]) = kv8;
#pragma unroll
        for (int j = 0; j < 8; ++j) vt_lds[dg + j][kvr] = vv8[j];
      }
    }
    __syncthreads();

    // ---- S = Q K^T over this tile (4 ntiles of 16 kv) ----
    f32x4 s_acc[NTILES];
#pragma unroll
    for (int nt = 0; nt < NTILES; ++nt) {
      f32x4 acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < KSTEPS_QK; ++ks) {
        bf16x8 b =
            *reinterpret_cast<const bf16x8*>(&k_lds[nt * 16 + lo][ks * 32 + hi * 8]);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(q_frag[ks], b, acc, 0,
                                                      0, 0);
      }
      s_acc[nt] = acc;
    }

    // ---- mask + scale; rowwise max ----
    float rowmax[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) rowmax[r] = -INFINITY;
#pragma unroll
    for (int nt = 0; nt < NTILES; ++nt) {
      int kv_idx = kvb + nt * 16 + lo;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int q_idx = qb * ATTN_BLOCK_Q + wave * 16 + hi * 4 + r;
        float sv = s_acc[nt][r] * scale;
        bool masked = (kv_idx >= Skv) || (causal && kv_idx > q_idx) ||
                      (q_idx >= S);
        sv = masked ? -INFINITY : sv;
        s_acc[nt][r] = sv;
        rowmax[r] = fmaxf(rowmax[r], sv);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        rowmax[r] = fmaxf(rowmax[r], __shfl_xor(rowmax[r], off, 16));
    }

    // ---- online rescale; P = exp(S - m_new); row sums ----
    float rowsum[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float m_new = fmaxf(m_state[r], rowmax[r]);
      float alpha = (m_state[r] == -INFINITY) ? 0.f : __expf(m_state[r] - m_new);
      m_state[r] = m_new;
      l_state[r] *= alpha;
#pragma unroll
      for (int dt = 0; dt < DTILES; ++dt) o_acc[dt][r] *= alpha;
      rowsum[r] = 0.f;
    }
#pragma unroll
    for (int nt = 0; nt < NTILES; ++nt) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = (s_acc[nt][r] == -INFINITY)
                      ? 0.f
                      : __expf(s_acc[nt][r] - m_state[r]);
        s_acc[nt][r] = p;
        rowsum[r] += p;
        // write P to the wave's LDS tile for re-fragmentation
        p_lds[wave][hi * 4 + r][nt * 16 + lo] = f2bf(p);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int off = 1; off < 16; off <<= 1)
        rowsum[r] += __shfl_xor(rowsum[r], off, 16);
      l_state[r] += rowsum[r];
    }

    // ---- O += P V  (A = P from LDS, B = Vt from LDS) ----
    // (p_lds write->read is wave-local; compiler inserts the lgkmcnt wait)
#pragma unroll
    for (int ks = 0; ks < ATTN_BLOCK_K / 32; ++ks) {
      bf16x8 a =
          *reinterpret_cast<const bf16x8*>(&p_lds[wave][lo][ks * 32 + hi * 8]);
#pragma unroll
      for (int dt = 0; dt < DTILES; ++dt) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &vt_lds[dt * 16 + lo][ks * 32 + hi * 8]);
        o_acc[dt] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, o_acc[dt], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: normalize, store O and lse ----
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int q_idx = qb * ATTN_BLOCK_Q + wave * 16 + hi * 4 + r;
    if (q_idx >= S) continue;
    float inv_l = (l_state[r] > 0.f) ? 1.0f / l_state[r] : 0.f;
#pragma unroll
    for (int dt = 0; dt < DTILES; ++dt) {
      int col = dt * 16 + lo;
      if (col < D)
        op[(int64_t)q_idx * st.os + col] = f2bf(o_acc[dt][r] * inv_l);
    }
    if (lo == 0 && lse_out != nullptr)
      lse_out[(int64_t)bh * S + q_idx] =
          (l_state[r] > 0.f) ? m_state[r] + __logf(l_state[r]) : -INFINITY;
  }
}

extern "C" {

hipError_t launch_attn_fwd(const void* q, const void* k, const void* v,
                           void* o, float* lse, int64_t B, int64_t H,
                           int64_t S, int64_t Skv, int64_t D, float scale,
                           int causal, const int64_t* strides,
                           hipStream_t stream) {
  dim3 grid((uint32_t)ceil_div(S, ATTN_BLOCK_Q), (uint32_t)(B * H));
  dim3 block(ATTN_THREADS);
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
    attn_fwd_kernel<64><<<grid, block, 0, stream>>>(
        (const short*)q, (const short*)k, (const short*)v, (short*)o, lse,
        (int)H, (int)S, (int)Skv, (int)D, scale, causal, st);
  } else if (D <= 96) {
    attn_fwd_kernel<96><<<grid, block, 0, stream>>>(
        (const short*)q, (const short*)k, (const short*)v, (short*)o, lse,
        (int)H, (int)S, (int)Skv, (int)D, scale, causal, st);
  } else if (D <= 128) {
    attn_fwd_kernel<128><<<grid, block, 0, stream>>>(
        (const short*)q, (const short*)k, (const short*)v, (short*)o, lse,
        (int)H, (int)S, (int)Skv, (int)D, scale, causal, st);
  } else {
    return hipErrorInvalidValue;
  }
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
  // one wave per row
  int64_t row = (int64_t)blockIdx.x * (blockDim.x >> 6) + (threadIdx.x >> 6);
  if (row >= rows) return;
  int lane = threadIdx.x & 63;
  int64_t sidx = row % S, h = (row / S) % H, b = row / ((int64_t)S * H);
  const short* dr = dout + b * dob + h * doh + sidx * dos;
  const short* orow = o + b * ob + h * oh + sidx * os;
  float s = 0.f;
  for (int i = lane * 4; i < D; i += 64 * 4) {
    bf16x4 dv = *reinterpret_cast<const bf16x4*>(dr + i);
    bf16x4 ov = *reinterpret_cast<const bf16x4*>(orow + i);
#pragma unroll
    for (int j = 0; j < 4; ++j) s += bf2f(dv[j]) * bf2f(ov[j]);
  }
  s = wave_reduce_sum(s);
  if (lane == 0) delta[row] = s;
}

// ---------------------------------------------------------------- dq kernel
// Block: 4 waves, 64 q rows; loops kv blocks of 64.
//   S = Q K^T          (A=Q regs, B=K_lds row-major)
//   P = exp(S*? - lse) (lse already includes the scale from fwd)
//   dP = dO V^T        (A=dO regs, B=V_lds row-major)
//   dS = P*(dP-delta)*scale
//   dQ += dS K         (A=dS via p_lds, B=Kt_lds transposed)
template <int Dp>
__global__ __launch_bounds__(ATTN_THREADS) void attn_bwd_dq_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dq, int H, int S, int Skv, int D, float scale,
    int causal, AttnBwdStrides st) {
  constexpr int KD = Dp / 32;          // k-steps over head dim
  constexpr int NT = 4;                // 64 kv per tile
  constexpr int DT = Dp / 16;

  const int qb = blockIdx.x;
  const int bh = blockIdx.y;
  const int batch = bh / H, head = bh % H;
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int lo = lane & 15, hi = lane >> 4;

  const short* qp = q + batch * st.qb + head * st.qh;
  const short* kp = k + batch * st.kb + head * st.kh;
  const short* vp = v + batch * st.vb + head * st.vh;
  const short* dop = dout + batch * st.dob + head * st.doh;
  short* dqp = dq + batch * st.dqb + head * st.dqh;
  const int q_row0 = qb * 64 + wave * 16;

  constexpr int LP = 8;  // bank-conflict row padding (see fwd kernel)
  __shared__ short k_lds[64][Dp + LP];
  __shared__ short kt_lds[Dp][64 + LP];
  __shared__ short v_lds[64][Dp + LP];
  __shared__ short p_lds[4][16][64 + LP];

  // Q and dO fragments in registers (A-operand: m=lo, k=hi*8+j)
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
  // per-row lse/delta (rows hi*4+r)
  float lse_r[4], delta_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int qi = q_row0 + hi * 4 + r;
    lse_r[r] = (qi < S) ? lse[(int64_t)bh * S + qi] : 0.f;
    delta_r[r] = (qi < S) ? delta[(int64_t)bh * S + qi] : 0.f;
  }

  f32x4 dq_acc[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) dq_acc[dt] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int kv_limit = causal ? min(Skv, qb * 64 + 64) : Skv;
  for (int kvb = 0; kvb < kv_limit; kvb += 64) {
    __syncthreads();
    {  // stage K (row + transposed) and V
      constexpr int GPR = Dp / 8;
      for (int t = threadIdx.x; t < 64 * GPR; t += ATTN_THREADS) {
        int kvr = t / GPR, dg = (t % GPR) * 8;
        bf16x8 kv8 = {0, 0, 0, 0, 0, 0, 0, 0};
        bf16x8 vv8 = {0, 0, 0, 0, 0, 0, 0, 0};
        int src = kvb + kvr;
        if (src < Skv && dg + 8 <= D) {
          kv8 = *reinterpret_cast<const bf16x8*>(kp + (int64_t)src * st.ks + dg);
          vv8 = *reinterpret_cast<const bf16x8*>(vp + (int64_t)src * st.vs + dg);
        }
        *reinterpret_cast<bf16x8*>(&k_lds[kvr][dg]) = kv8;
        *reinterpret_cast<bf16x8*>(&v_lds[kvr][dg]) = vv8;
#pragma unroll
        for (int j = 0; j < 8; ++j) kt_lds[dg + j][kvr] = kv8[j];
      }
    }
    __syncthreads();

    // S and dP tiles
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

    // dS = P * (dP - delta) * scale, written to p_lds as A-operand
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
      int kv_idx = kvb + nt * 16 + lo;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int q_idx = qb * 64 + wave * 16 + hi * 4 + r;
        bool masked = (kv_idx >= Skv) || (causal && kv_idx > q_idx) ||
                      (q_idx >= S);
        float p = masked ? 0.f
                         : __expf(s_acc[nt][r] * scale - lse_r[r]);
        float ds = p * (dp_acc[nt][r] - delta_r[r]) * scale;
        p_lds[wave][hi * 4 + r][nt * 16 + lo] = f2bf(ds);
      }
    }

    // dQ += dS @ K   (A = dS from p_lds, B = Kt_lds)
#pragma unroll
    for (int ks = 0; ks < 64 / 32; ++ks) {
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &p_lds[wave][lo][ks * 32 + hi * 8]);
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &kt_lds[dt * 16 + lo][ks * 32 + hi * 8]);
        dq_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b,
                                                             dq_acc[dt], 0,
                                                             0, 0);
      }
    }
  }

  // store dQ (C-layout scatter)
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int q_idx = qb * 64 + wave * 16 + hi * 4 + r;
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
template <int Dp>
__global__ __launch_bounds__(ATTN_THREADS) void attn_bwd_dkv_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ delta,
    short* __restrict__ dk, short* __restrict__ dv, int H, int S, int Skv,
    int D, float scale, int causal, AttnBwdStrides st) {
  constexpr int KD = Dp / 32;
  constexpr int NT = 4;  // 64 q per tile
  constexpr int DT = Dp / 16;

  const int kvb_idx = blockIdx.x;
  const int bh = blockIdx.y;
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
  const int kv_row0 = kvb_idx * 64 + wave * 16;

  constexpr int LP = 8;  // bank-conflict row padding (see fwd kernel)
  __shared__ short q_lds[64][Dp + LP];
  __shared__ short qt_lds[Dp][64 + LP];
  __shared__ short do_lds[64][Dp + LP];
  __shared__ short dot_lds[Dp][64 + LP];
  __shared__ short p_lds[4][16][64 + LP];
  __shared__ float lse_lds[64];
  __shared__ float delta_lds[64];

  // K and V fragments in registers (A-operand)
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

  const int q_start = causal ? (kvb_idx * 64 / 64) * 64 : 0;
  for (int qb = q_start; qb < S; qb += 64) {
    __syncthreads();
    {  // stage Q, dO (row + transposed) and lse/delta
      constexpr int GPR = Dp / 8;
      for (int t = threadIdx.x; t < 64 * GPR; t += ATTN_THREADS) {
        int qr = t / GPR, dg = (t % GPR) * 8;
        bf16x8 q8 = {0, 0, 0, 0, 0, 0, 0, 0};
        bf16x8 d8 = {0, 0, 0, 0, 0, 0, 0, 0};
        int src = qb + qr;
        if (src < S && dg + 8 <= D) {
          q8 = *reinterpret_cast<const bf16x8*>(qp + (int64_t)src * st.qs + dg);
          d8 = *reinterpret_cast<const bf16x8*>(dop + (int64_t)src * st.dos + dg);
        }
        *reinterpret_cast<bf16x8*>(&q_lds[qr][dg]) = q8;
        *reinterpret_cast<bf16x8*>(&do_lds[qr][dg]) = d8;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          qt_lds[dg + j][qr] = q8[j];
          dot_lds[dg + j][qr] = d8[j];
        }
      }
      for (int t = threadIdx.x; t < 64; t += ATTN_THREADS) {
        int src = qb + t;
        lse_lds[t] = (src < S) ? lse[(int64_t)bh * S + src] : 0.f;
        delta_lds[t] = (src < S) ? delta[(int64_t)bh * S + src] : 0.f;
      }
    }
    __syncthreads();

    // S^T and dP^T tiles (rows = kv, cols = q)
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

    // P^T -> p_lds for dV; then dS^T -> p_lds for dK (two passes over the
    // same per-wave buffer, separated by the MFMA consumption)
    float pt[NT][4], dst[NT][4];
#pragma unroll
    for (int nt = 0; nt < NT; ++nt) {
      int q_idx = qb + nt * 16 + lo;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int kv_idx = kvb_idx * 64 + wave * 16 + hi * 4 + r;
        bool masked = (q_idx >= S) || (kv_idx >= Skv) ||
                      (causal && kv_idx > q_idx);
        float p = masked ? 0.f
                         : __expf(st_acc[nt][r] * scale - lse_lds[nt * 16 + lo]);
        pt[nt][r] = p;
        dst[nt][r] = p * (dpt_acc[nt][r] - delta_lds[nt * 16 + lo]) * scale;
      }
    }

    // dV += P^T @ dO (A = P^T, B = dOt)
#pragma unroll
    for (int nt = 0; nt < NT; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        p_lds[wave][hi * 4 + r][nt * 16 + lo] = f2bf(pt[nt][r]);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &p_lds[wave][lo][ks * 32 + hi * 8]);
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &dot_lds[dt * 16 + lo][ks * 32 + hi * 8]);
        dv_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b,
                                                             dv_acc[dt], 0,
                                                             0, 0);
      }
    }

    // dK += dS^T @ Q (A = dS^T, B = Qt)
#pragma unroll
    for (int nt = 0; nt < NT; ++nt)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        p_lds[wave][hi * 4 + r][nt * 16 + lo] = f2bf(dst[nt][r]);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8 a = *reinterpret_cast<const bf16x8*>(
          &p_lds[wave][lo][ks * 32 + hi * 8]);
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        bf16x8 b = *reinterpret_cast<const bf16x8*>(
            &qt_lds[dt * 16 + lo][ks * 32 + hi * 8]);
        dk_acc[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b,
                                                             dk_acc[dt], 0,
                                                             0, 0);
      }
    }
  }

  // store dK, dV
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int kv_idx = kvb_idx * 64 + wave * 16 + hi * 4 + r;
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

hipError_t launch_attn_bwd(const void* q, const void* k, const void* v,
                           const void* o, const void* dout,
                           const float* lse, float* delta_ws, void* dq,
                           void* dk, void* dv, int64_t B, int64_t H,
                           int64_t S, int64_t Skv, int64_t D, float scale,
                           int causal, const int64_t* strides,
                           hipStream_t stream) {
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
    dim3 grid((uint32_t)ceil_div(rows, 4));
    attn_bwd_preprocess_kernel<<<grid, block, 0, stream>>>(
        (const short*)dout, (const short*)o, delta_ws, rows, (int)H, (int)S,
        (int)D, st.dob, st.doh, st.dos, p[21], p[22], p[23]);
  }
  dim3 block(ATTN_THREADS);
  dim3 grid_q((uint32_t)ceil_div(S, 64), (uint32_t)(B * H));
  dim3 grid_kv((uint32_t)ceil_div(Skv, 64), (uint32_t)(B * H));
#define LAUNCH_BWD(DP)                                                       \
  do {                                                                       \
    attn_bwd_dq_kernel<DP><<<grid_q, block, 0, stream>>>(                    \
        (const short*)q, (const short*)k, (const short*)v,                   \
        (const short*)dout, lse, delta_ws, (short*)dq, (int)H, (int)S,       \
        (int)Skv, (int)D, scale, causal, st);                                \
    attn_bwd_dkv_kernel<DP><<<grid_kv, block, 0, stream>>>(                  \
        (const short*)q, (const short*)k, (const short*)v,                   \
        (const short*)dout, lse, delta_ws, (short*)dk, (short*)dv, (int)H,   \
        (int)S, (int)Skv, (int)D, scale, causal, st);                        \
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
  return hipGetLastError();
}

}  // extern "C"
