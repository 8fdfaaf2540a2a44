// Fused softmax-cross-entropy over the vocab dim (gfx950).
//
// logits [N, V] bf16 (V = 51200 for the GPT ladder), targets [N] int64.
// Forward: one block per row, single online max+sumexp pass (never
// materializes a softmax tensor).  Backward: dlogits = (softmax - onehot)
// * dloss recomputed from (logits, lse).
#include "common.h"

#define CE_BLOCK 256

__global__ void ce_fwd_kernel(const short* __restrict__ logits,
                              const int64_t* __restrict__ targets,
                              float* __restrict__ loss,
                              float* __restrict__ lse_out, int V) {
  const int row = blockIdx.x;
  const short* lr = logits + (int64_t)row * V;
  __shared__ float red[CE_BLOCK / 64];

  // online max + sum(exp(x - m))
  float m = -INFINITY, s = 0.f;
  for (int i = threadIdx.x * 8; i < V; i += CE_BLOCK * 8) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(lr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v[j]);
      if (f > m) {
        s *= __expf(m - f);
        m = f;
      }
      s += __expf(f - m);
    }
  }
  // combine across block: need global max first
  float gm = block_reduce_max(m, red);
  s *= __expf(m - gm);
  float gs = block_reduce_sum(s, red);
  float lse = gm + __logf(gs);
  if (threadIdx.x == 0) {
    int64_t t = targets[row];
    loss[row] = lse - bf2f(lr[t]);
    lse_out[row] = lse;
  }
}

__global__ void ce_bwd_kernel(const float* __restrict__ dloss,
                              const short* __restrict__ logits,
                              const int64_t* __restrict__ targets,
                              const float* __restrict__ lse,
                              short* __restrict__ dlogits, int V) {
  const int row = blockIdx.x;
  const short* lr = logits + (int64_t)row * V;
  short* dr = dlogits + (int64_t)row * V;
  const float l = lse[row];
  const float dl = dloss[row];
  const int64_t t = targets[row];
  for (int i = threadIdx.x * 8; i < V; i += CE_BLOCK * 8) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(lr + i);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float p = __expf(bf2f(v[j]) - l);
      if ((int64_t)(i + j) == t) p -= 1.0f;
      o[j] = f2bf(p * dl);
    }
    *reinterpret_cast<bf16x8*>(dr + i) = o;
  }
}

extern "C" {

hipError_t launch_ce_fwd(const void* logits, const int64_t* targets,
                         float* loss, float* lse, int64_t N, int64_t V,
                         hipStream_t stream) {
  ce_fwd_kernel<<<dim3((uint32_t)N), dim3(CE_BLOCK), 0, stream>>>(
      (const short*)logits, targets, loss, lse, (int)V);
  return hipGetLastError();
}

hipError_t launch_ce_bwd(const float* dloss, const void* logits,
                         const int64_t* targets, const float* lse,
                         void* dlogits, int64_t N, int64_t V,
                         hipStream_t stream) {
  ce_bwd_kernel<<<dim3((uint32_t)N), dim3(CE_BLOCK), 0, stream>>>(
      dloss, (const short*)logits, targets, lse, (short*)dlogits, (int)V);
  return hipGetLastError();
}

}  // extern "C"
