// Common helpers for the gfx950 (CDNA4/MI355X) kernel library.
//
// Hand-written HIP, wave64, MFMA bf16. No CUDA shims, no hipify.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#include <cstdint>

#define WAVE_SIZE 64

using bf16 = __hip_bfloat16;

// vectorized load types (G13: always vectorize bf16 loads — 8-16 B/lane)
typedef short bf16x8 __attribute__((ext_vector_type(8)));   // 16 B
typedef short bf16x4 __attribute__((ext_vector_type(4)));   // 8 B
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

__device__ __forceinline__ float bf2f(short raw) {
  union {
    float f;
    uint32_t u;
  } cvt;
  cvt.u = (uint32_t)(uint16_t)raw << 16;
  return cvt.f;
}

__device__ __forceinline__ short f2bf(float f) {
  union {
    float f;
    uint32_t u;
  } cvt;
  cvt.f = f;
  // round-to-nearest-even like hardware
  uint32_t lsb = (cvt.u >> 16) & 1;
  cvt.u += 0x7fff + lsb;
  return (short)(cvt.u >> 16);
}

// ---------------- wave reductions (64 lanes) ----------------
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, 64);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    v = fmaxf(v, __shfl_xor(v, off, 64));
  return v;
}

// Block reduction via LDS (blockDim.x threads, <= 1024).
// `buf` must hold blockDim.x/64 floats.
__device__ __forceinline__ float block_reduce_sum(float v, float* buf) {
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  int nwaves = blockDim.x >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) buf[wave] = v;
  __syncthreads();
  float r = (lane < nwaves) ? buf[lane] : 0.0f;
  r = wave_reduce_sum(r);  // small over-reduce; lanes>=nwaves contribute 0
  __syncthreads();
  return r;
}

__device__ __forceinline__ float block_reduce_max(float v, float* buf) {
  int lane = threadIdx.x & 63;
  int wave = threadIdx.x >> 6;
  int nwaves = blockDim.x >> 6;
  v = wave_reduce_max(v);
  if (lane == 0) buf[wave] = v;
  __syncthreads();
  float r = (lane < nwaves) ? buf[lane] : -INFINITY;
  r = wave_reduce_max(r);
  __syncthreads();
  return r;
}

__host__ __forceinline__ int64_t ceil_div(int64_t a, int64_t b) {
  return (a + b - 1) / b;
}
