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

  __shared__ short k_lds[ATTN_BLOCK_K][Dp];
  __shared__ short vt_lds[Dp][ATTN_BLOCK_K];
  __shared__ short p_lds[4][16][ATTN_BLOCK_K];  // per-wave P tile

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
