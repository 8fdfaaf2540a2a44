// Fused bias + GeLU(tanh) forward/backward, bf16 (gfx950).
//
// The epilogue of the MLP up-projection GEMM (reference gets this from XLA
// fusion; SURVEY.md §2.3 N13 "GEMM+bias+GeLU epilogue").  Memory-bound:
// bf16x8 vector loads, grid-stride, dbias via deterministic striped
// partials + column sum.
#include "common.h"

// tanh-GeLU via the sigmoid identity: 0.5x(1+tanh(z)) == x*sigmoid(2z),
// evaluated with ONE hardware __expf instead of libm tanhf (the tanh
// path ran the elementwise kernels 1.7-1.9x off the HBM roofline —
// VALU-bound on the polynomial tanh expansion).  Same function, only
// the evaluation differs; bf16 outputs agree to rounding.
__device__ __forceinline__ float gelu_sigmoid(float x) {
  const float k2 = 2.0f * 0.7978845608028654f;
  float z2 = k2 * (x + 0.044715f * x * x * x);
  return 1.0f / (1.0f + __expf(-z2));
}

__device__ __forceinline__ float gelu_f(float x) {
  return x * gelu_sigmoid(x);
}

__device__ __forceinline__ float gelu_grad_f(float x) {
  const float k = 0.7978845608028654f;
  float x2 = x * x;
  float sg = gelu_sigmoid(x);
  // t = 2*sg - 1;  1 - t^2 = 4*sg*(1-sg)
  return sg + 2.0f * x * sg * (1.0f - sg) * k *
                  (1.0f + 3.f * 0.044715f * x2);
}

// x [N, F] bf16, bias [F] -> y [N, F]
__global__ void bias_gelu_fwd_kernel(const short* __restrict__ x,
                                     const short* __restrict__ bias,
                                     short* __restrict__ y, int64_t total,
                                     int F) {
  int64_t idx = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (; idx < total; idx += stride) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(x + idx);
    bf16x8 bv = *reinterpret_cast<const bf16x8*>(bias + (idx % F));
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o[j] = f2bf(gelu_f(bf2f(v[j]) + bf2f(bv[j])));
    *reinterpret_cast<bf16x8*>(y + idx) = o;
  }
}

__global__ void bias_gelu_bwd_dx_kernel(const short* __restrict__ dy,
                                        const short* __restrict__ x,
                                        const short* __restrict__ bias,
                                        short* __restrict__ dx, int64_t total,
                                        int F) {
  int64_t idx = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  for (; idx < total; idx += stride) {
    bf16x8 dv = *reinterpret_cast<const bf16x8*>(dy + idx);
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(x + idx);
    bf16x8 bv = *reinterpret_cast<const bf16x8*>(bias + (idx % F));
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float z = bf2f(xv[j]) + bf2f(bv[j]);
      o[j] = f2bf(bf2f(dv[j]) * gelu_grad_f(z));
    }
    *reinterpret_cast<bf16x8*>(dx + idx) = o;
  }
}

// dbias partials: stripe p sums its row range of dx over an 8-column
// group per thread (bf16x8 loads — the scalar version was issue-bound
// at ~6x off the HBM roofline).
__global__ void bias_grad_partial_kernel(const short* __restrict__ dx,
                                         float* __restrict__ part, int64_t N,
                                         int F) {
  const int col8 = (blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const int p = blockIdx.y;
  const int P = gridDim.y;
  const int64_t r0 = (int64_t)p * N / P, r1 = (int64_t)(p + 1) * N / P;
  if (col8 + 8 <= F) {
    float s[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    for (int64_t row = r0; row < r1; ++row) {
      bf16x8 v = *reinterpret_cast<const bf16x8*>(dx + row * F + col8);
#pragma unroll
      for (int j = 0; j < 8; ++j) s[j] += bf2f(v[j]);
    }
#pragma unroll
    for (int j = 0; j < 8; ++j) part[(int64_t)p * F + col8 + j] = s[j];
  } else if (col8 < F) {
    for (int c = col8; c < F; ++c) {
      float s = 0.f;
      for (int64_t row = r0; row < r1; ++row) s += bf2f(dx[row * F + c]);
      part[(int64_t)p * F + c] = s;
    }
  }
}

__global__ void bias_colsum_kernel(const float* __restrict__ part,
                                   float* __restrict__ out, int P, int F) {
  int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= F) return;
  float s = 0.f;
  for (int p = 0; p < P; ++p) s += part[(int64_t)p * F + col];
  out[col] = s;
}

extern "C" {

hipError_t launch_bias_gelu_fwd(const void* x, const void* bias, void* y,
                                int64_t N, int64_t F, hipStream_t stream) {
  int64_t total = N * F;
  int blocks = (int)min((int64_t)2048, ceil_div(total, 256 * 8));
  bias_gelu_fwd_kernel<<<dim3(blocks), dim3(256), 0, stream>>>(
      (const short*)x, (const short*)bias, (short*)y, total, (int)F);
  return hipGetLastError();
}

hipError_t launch_colsum_partial(const void* t, float* part, int64_t N,
                                 int64_t F, int P, hipStream_t stream) {
  dim3 grid((uint32_t)ceil_div(F, 256 * 8), P);
  bias_grad_partial_kernel<<<grid, dim3(256), 0, stream>>>(
      (const short*)t, part, N, (int)F);
  return hipGetLastError();
}

hipError_t launch_bias_gelu_bwd(const void* dy, const void* x,
                                const void* bias, void* dx, float* db_part,
                                float* db, int64_t N, int64_t F, int P,
                                hipStream_t stream) {
  int64_t total = N * F;
  int blocks = (int)min((int64_t)2048, ceil_div(total, 256 * 8));
  bias_gelu_bwd_dx_kernel<<<dim3(blocks), dim3(256), 0, stream>>>(
      (const short*)dy, (const short*)x, (const short*)bias, (short*)dx,
      total, (int)F);
  dim3 grid((uint32_t)ceil_div(F, 256 * 8), P);
  bias_grad_partial_kernel<<<grid, dim3(256), 0, stream>>>(
      (const short*)dx, db_part, N, (int)F);
  // final P-row reduction handled by the caller (at::sum)
  return hipGetLastError();
}

}  // extern "C"
