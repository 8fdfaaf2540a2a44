// Skinny-M GEMM (decode GEMV) for gfx950: y[M,N] = x[M,K] @ W[N,K]^T,
// M <= 64 (one decode step of a serving batch).
//
// Why: hipBLASLt's bf16 matmul runs these shapes at ~2.8 TB/s (19 us
// for 5120x5120 at M=16, measured tools/fp8_decode_probe.py) where the
// weight-streaming floor is ~6.5 us — decode is weight-bandwidth-bound
// and the library kernel leaves ~3x on the table.  fp8 never wins there
// (no skinny fp8 path below M=128).
//
// Design (lane-per-row over a packed layout):
//   - Weights are PRE-PACKED once at serving load into Wp[K/8][N][8]
//     (k-major 8-element groups): for a fixed k-group, 64 consecutive
//     rows' 16-byte segments are CONTIGUOUS, so a wave of 64 lanes
//     (lane = row) issues one fully-coalesced 1-KB load per k-round.
//     The natural [N,K] layout would stride lanes K*2 bytes apart.
//   - The workgroup's x slice [MT, rounds*8] is staged in LDS once
//     (transposed to pair-major so the inner loop reads ds_read_b64 at
//     wave-uniform addresses — broadcast, conflict-free).  v1 read x
//     from global per round per m and was vector-load bound.
//   - The k-round loop is UNROLLED 8x so 8 weight loads (128 B/lane)
//     are in flight before the dependent dot2 chain — v1 had exactly
//     one load outstanding and ran at HBM latency (374 ns/round), not
//     bandwidth.
//   - Each lane accumulates acc[m] (<= MT VGPRs) for ITS row — no
//     cross-lane reduction.  Split-K workgroups STORE fp32 partials to
//     part[S, MT, N] (no zeroing, no atomics) and a finalize kernel
//     reduces the S slices + adds bias + casts to bf16 in fixed order —
//     2 launches total and deterministic (the atomicAdd variant needed
//     an at::zeros + cast + bias_add per call: ~3 extra launches per
//     linear, which erased the GEMV win at 320 linears/decode-token on
//     OPT-66B).
//   - fp8 (e4m3) weight variant reads HALF the bytes and dequantizes
//     in-kernel via the packed hardware converter; x stays bf16.
//
// Grid: (N/64, S); block = 64 threads (1 wave).  The launcher picks S
// (see ops/__init__.py _skinny_splits) to fill the 256 CUs AND bound
// the LDS slice; K % (8*S) == 0, N % 64 == 0, rounds % 8 == 0.
#include "common.h"

typedef unsigned char u8x8 __attribute__((ext_vector_type(8)));
typedef float f32x2_t __attribute__((ext_vector_type(2)));
typedef short bf16x2 __attribute__((ext_vector_type(2)));
typedef int i32x2 __attribute__((ext_vector_type(2)));

#define UNR 8  // k-rounds per software-pipelined block

// dot of a packed bf16 pair with accumulate (v_dot2_f32_bf16)
__device__ __forceinline__ float dot2_bf16(bf16x2 a, bf16x2 b, float acc) {
  return __builtin_amdgcn_fdot2_f32_bf16(a, b, acc, false);
}

__device__ __forceinline__ bf16x2 mk2(short a, short b) {
  bf16x2 r;
  r[0] = a;
  r[1] = b;
  return r;
}

__device__ __forceinline__ bf16x2 pair_lo(int v) {
  bf16x2 r;
  r[0] = (short)(v & 0xffff);
  r[1] = (short)((unsigned int)v >> 16);
  return r;
}

// Stage x[0:MT, kg0*8 : (kg0+rounds)*8] into LDS, transposed to
// half-round-major: xs[(r*2 + h) * MT + m] = i32x2 holding k-pairs
// (4h, 4h+1) of round r for row m.  Coalesced 16-B global reads.
template <int MT>
__device__ __forceinline__ void stage_x(const short* __restrict__ x,
                                        i32x2* xs, int64_t K, int kg0,
                                        int rounds) {
  for (int i = threadIdx.x; i < MT * rounds; i += 64) {
    const int m = i / rounds, r = i - (i / rounds) * rounds;
    bf16x4 lo = *reinterpret_cast<const bf16x4*>(
        x + (int64_t)m * K + ((int64_t)(kg0 + r)) * 8);
    bf16x4 hi = *reinterpret_cast<const bf16x4*>(
        x + (int64_t)m * K + ((int64_t)(kg0 + r)) * 8 + 4);
    i32x2 a, b;
    a[0] = ((int)(unsigned short)lo[0]) | ((int)(unsigned short)lo[1] << 16);
    a[1] = ((int)(unsigned short)lo[2]) | ((int)(unsigned short)lo[3] << 16);
    b[0] = ((int)(unsigned short)hi[0]) | ((int)(unsigned short)hi[1] << 16);
    b[1] = ((int)(unsigned short)hi[2]) | ((int)(unsigned short)hi[3] << 16);
    xs[(r * 2 + 0) * MT + m] = a;
    xs[(r * 2 + 1) * MT + m] = b;
  }
  __syncthreads();
}

template <int MT>
__launch_bounds__(64, 1)  // 64-thread blocks; let big-MT tiers use the
                          // full VGPR file instead of spilling
__global__ void skinny_gemm_bf16_kernel(
    const short* __restrict__ wp,   // [K/8, N, 8] packed bf16
    const short* __restrict__ x,    // [MT, K] bf16 (padded rows zero)
    float* __restrict__ part,       // [S, MT, N] fp32 (overwritten)
    int64_t N, int64_t K, int rounds) {
  extern __shared__ i32x2 xs[];
  const int64_t row = (int64_t)blockIdx.x * 64 + threadIdx.x;
  const int kg0 = blockIdx.y * rounds;
  stage_x<MT>(x, xs, K, kg0, rounds);
  float acc[MT];
#pragma unroll
  for (int m = 0; m < MT; ++m) acc[m] = 0.f;
  const short* wrow = wp + ((int64_t)kg0 * N + row) * 8;
  const int64_t wstep = N * 8;
  for (int rb = 0; rb < rounds; rb += UNR) {
    bf16x8 w[UNR];
#pragma unroll
    for (int u = 0; u < UNR; ++u)
      w[u] = *reinterpret_cast<const bf16x8*>(wrow + (int64_t)u * wstep);
    wrow += (int64_t)UNR * wstep;
    // m-outer: the UNR*2 LDS reads for one m are independent — they
    // issue back-to-back and wait ONCE (the u-outer variant paid the
    // full ds_read latency per read); 4 accumulator chains keep the
    // 4*UNR dot2 out of one serial dependency.
#pragma unroll
    for (int m = 0; m < MT; ++m) {
      i32x2 a[UNR], b[UNR];
#pragma unroll
      for (int u = 0; u < UNR; ++u) {
        a[u] = xs[((rb + u) * 2 + 0) * MT + m];
        b[u] = xs[((rb + u) * 2 + 1) * MT + m];
      }
      float s0 = acc[m], s1 = 0.f, s2 = 0.f, s3 = 0.f;
#pragma unroll
      for (int u = 0; u < UNR; ++u) {
        s0 = dot2_bf16(mk2(w[u][0], w[u][1]), pair_lo(a[u][0]), s0);
        s1 = dot2_bf16(mk2(w[u][2], w[u][3]), pair_lo(a[u][1]), s1);
        s2 = dot2_bf16(mk2(w[u][4], w[u][5]), pair_lo(b[u][0]), s2);
        s3 = dot2_bf16(mk2(w[u][6], w[u][7]), pair_lo(b[u][1]), s3);
      }
      acc[m] = (s0 + s1) + (s2 + s3);
    }
  }
  float* out = part + ((int64_t)blockIdx.y * MT) * N + row;
#pragma unroll
  for (int m = 0; m < MT; ++m) out[(int64_t)m * N] = acc[m];
}

// fp8 weights: Wp[K/8][N][8] e4m3 bytes, 8-B loads; dequant scale
// folded in once at the end (per-tensor).
template <int MT>
__launch_bounds__(64, 1)
__global__ void skinny_gemm_fp8_kernel(
    const unsigned char* __restrict__ wp,  // [K/8, N, 8] packed e4m3
    const short* __restrict__ x,           // [MT, K] bf16
    float* __restrict__ part,              // [S, MT, N] fp32 (overwritten)
    const float* __restrict__ wscale,      // [1]
    int64_t N, int64_t K, int rounds) {
  extern __shared__ i32x2 xs[];
  const int64_t row = (int64_t)blockIdx.x * 64 + threadIdx.x;
  const int kg0 = blockIdx.y * rounds;
  stage_x<MT>(x, xs, K, kg0, rounds);
  float acc[MT];
#pragma unroll
  for (int m = 0; m < MT; ++m) acc[m] = 0.f;
  const unsigned char* wrow = wp + ((int64_t)kg0 * N + row) * 8;
  const int64_t wstep = N * 8;
  for (int rb = 0; rb < rounds; rb += UNR) {
    u8x8 wb[UNR];
#pragma unroll
    for (int u = 0; u < UNR; ++u)
      wb[u] = *reinterpret_cast<const u8x8*>(wrow + (int64_t)u * wstep);
    wrow += (int64_t)UNR * wstep;
    // convert the whole block's weights to bf16 pairs up front
    bf16x8 w[UNR];
#pragma unroll
    for (int u = 0; u < UNR; ++u) {
      unsigned int lo = (unsigned int)wb[u][0] |
                        ((unsigned int)wb[u][1] << 8) |
                        ((unsigned int)wb[u][2] << 16) |
                        ((unsigned int)wb[u][3] << 24);
      unsigned int hi = (unsigned int)wb[u][4] |
                        ((unsigned int)wb[u][5] << 8) |
                        ((unsigned int)wb[u][6] << 16) |
                        ((unsigned int)wb[u][7] << 24);
      f32x2_t p0 = __builtin_amdgcn_cvt_pk_f32_fp8(lo, false);
      f32x2_t p1 = __builtin_amdgcn_cvt_pk_f32_fp8(lo, true);
      f32x2_t p2 = __builtin_amdgcn_cvt_pk_f32_fp8(hi, false);
      f32x2_t p3 = __builtin_amdgcn_cvt_pk_f32_fp8(hi, true);
      w[u][0] = f2bf(p0[0]); w[u][1] = f2bf(p0[1]);
      w[u][2] = f2bf(p1[0]); w[u][3] = f2bf(p1[1]);
      w[u][4] = f2bf(p2[0]); w[u][5] = f2bf(p2[1]);
      w[u][6] = f2bf(p3[0]); w[u][7] = f2bf(p3[1]);
    }
#pragma unroll
    for (int m = 0; m < MT; ++m) {
      i32x2 a[UNR], b[UNR];
#pragma unroll
      for (int u = 0; u < UNR; ++u) {
        a[u] = xs[((rb + u) * 2 + 0) * MT + m];
        b[u] = xs[((rb + u) * 2 + 1) * MT + m];
      }
      float s0 = acc[m], s1 = 0.f, s2 = 0.f, s3 = 0.f;
#pragma unroll
      for (int u = 0; u < UNR; ++u) {
        s0 = dot2_bf16(mk2(w[u][0], w[u][1]), pair_lo(a[u][0]), s0);
        s1 = dot2_bf16(mk2(w[u][2], w[u][3]), pair_lo(a[u][1]), s1);
        s2 = dot2_bf16(mk2(w[u][4], w[u][5]), pair_lo(b[u][0]), s2);
        s3 = dot2_bf16(mk2(w[u][6], w[u][7]), pair_lo(b[u][1]), s3);
      }
      acc[m] = (s0 + s1) + (s2 + s3);
    }
  }
  const float ws = *wscale;
  float* out = part + ((int64_t)blockIdx.y * MT) * N + row;
#pragma unroll
  for (int m = 0; m < MT; ++m) out[(int64_t)m * N] = acc[m] * ws;
}

// Reduce the S split slices in fixed order, add bias, cast to bf16.
__global__ void skinny_finalize_kernel(const float* __restrict__ part,
                                       const short* __restrict__ bias,
                                       short* __restrict__ y, int64_t MT,
                                       int64_t M, int64_t N, int splits) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= M * N) return;
  const int64_t m = i / N, n = i - m * N;
  float s = 0.f;
  for (int k = 0; k < splits; ++k) s += part[((int64_t)k * MT + m) * N + n];
  if (bias) s += bf2f(bias[n]);
  y[i] = f2bf(s);
}

extern "C" {

// dispatch over padded M tier; returns hipErrorInvalidValue on bad dims
hipError_t launch_skinny_gemm(const void* wp, const void* x, float* part,
                              const void* bias_or_null, void* y,
                              const float* wscale_or_null, int64_t M,
                              int64_t MT, int64_t N, int64_t K, int splits,
                              int fp8, hipStream_t stream) {
  if (N % 64 || K % 8 || (K / 8) % splits) return hipErrorInvalidValue;
  int rounds = (int)(K / 8 / splits);
  if (rounds % UNR) return hipErrorInvalidValue;
  dim3 grid((uint32_t)(N / 64), (uint32_t)splits);
  dim3 blk(64);
#define DISPATCH(MTC)                                                      \
  do {                                                                     \
    size_t lds = (size_t)rounds * 2 * MTC * sizeof(i32x2);                 \
    if (fp8)                                                               \
      skinny_gemm_fp8_kernel<MTC><<<grid, blk, lds, stream>>>(             \
          (const unsigned char*)wp, (const short*)x, part,                 \
          wscale_or_null, N, K, rounds);                                   \
    else                                                                   \
      skinny_gemm_bf16_kernel<MTC><<<grid, blk, lds, stream>>>(            \
          (const short*)wp, (const short*)x, part, N, K, rounds);          \
  } while (0)
  if (MT == 4) DISPATCH(4);
  else if (MT == 8) DISPATCH(8);
  else if (MT == 16) DISPATCH(16);
  else if (MT == 32) DISPATCH(32);
  else if (MT == 64) DISPATCH(64);
  else return hipErrorInvalidValue;
#undef DISPATCH
  int64_t total = M * N;
  int blocks = (int)((total + 255) / 256);
  skinny_finalize_kernel<<<dim3(blocks), dim3(256), 0, stream>>>(
      part, (const short*)bias_or_null, (short*)y, MT, M, N, splits);
  return hipGetLastError();
}

}  // extern "C"
