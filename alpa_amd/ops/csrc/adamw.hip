// Multi-tensor fused AdamW (gfx950).
//
// One kernel launch updates every parameter: a host-built chunk table maps
// each workgroup to (tensor_idx, offset).  fp32 math against fp32 moment
// state; params/grads may be bf16 or fp32.  grad_scale folds the
// 1/(num_microbatches * dp) division into the update (the reference folds
// the same division into its apply_grad jaxpr,
// shard_parallel/compile_executable.py:272).
#include "common.h"

#define ADAM_CHUNK 16384
#define ADAM_BLOCK 256

struct AdamTensorDesc {
  void* p;
  void* g;
  float* m;
  float* v;
  int64_t numel;
  int32_t is_bf16;  // dtype of p/g
  int32_t _pad;
};

struct AdamChunk {
  int32_t tensor_idx;
  int32_t chunk_idx;  // offset = chunk_idx * ADAM_CHUNK
};

__global__ void adamw_kernel(const AdamTensorDesc* __restrict__ descs,
                             const AdamChunk* __restrict__ chunks,
                             int num_chunks, float lr, float beta1,
                             float beta2, float eps, float weight_decay,
                             float grad_scale, float bc1, float bc2) {
  int cid = blockIdx.x;
  if (cid >= num_chunks) return;
  AdamChunk c = chunks[cid];
  AdamTensorDesc d = descs[c.tensor_idx];
  int64_t base = (int64_t)c.chunk_idx * ADAM_CHUNK;
  int64_t end = min(base + ADAM_CHUNK, d.numel);

  const float wd_factor = 1.0f - lr * weight_decay;
  for (int64_t i = base + threadIdx.x; i < end; i += ADAM_BLOCK) {
    float g, p;
    if (d.is_bf16) {
      g = bf2f(((const short*)d.g)[i]) * grad_scale;
      p = bf2f(((const short*)d.p)[i]);
    } else {
      g = ((const float*)d.g)[i] * grad_scale;
      p = ((const float*)d.p)[i];
    }
    float m = d.m[i] = d.m[i] * beta1 + g * (1.0f - beta1);
    float v = d.v[i] = d.v[i] * beta2 + g * g * (1.0f - beta2);
    p = p * wd_factor;
    p -= lr * (m / bc1) / (sqrtf(v / bc2) + eps);
    if (d.is_bf16)
      ((short*)d.p)[i] = f2bf(p);
    else
      ((float*)d.p)[i] = p;
  }
}

extern "C" {

hipError_t launch_adamw(const void* descs_dev, const void* chunks_dev,
                        int num_chunks, float lr, float beta1, float beta2,
                        float eps, float weight_decay, float grad_scale,
                        int step, hipStream_t stream) {
  float bc1 = 1.0f - powf(beta1, (float)step);
  float bc2 = 1.0f - powf(beta2, (float)step);
  adamw_kernel<<<dim3(num_chunks), dim3(ADAM_BLOCK), 0, stream>>>(
      (const AdamTensorDesc*)descs_dev, (const AdamChunk*)chunks_dev,
      num_chunks, lr, beta1, beta2, eps, weight_decay, grad_scale, bc1, bc2);
  return hipGetLastError();
}

}  // extern "C"
