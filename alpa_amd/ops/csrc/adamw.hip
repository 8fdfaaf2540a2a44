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
  // vectorized main loop: 8 elements/thread/iteration (G13 — the r1
  // scalar kernel ran 2x off the HBM roofline on 2-byte accesses);
  // torch allocations are 256B-aligned and chunk bases are multiples
  // of ADAM_CHUNK, so 16B vector access is aligned
  const int64_t n_full = (end - base) / 8 * 8;
  for (int64_t i = base + (int64_t)threadIdx.x * 8; i < base + n_full;
       i += (int64_t)ADAM_BLOCK * 8) {
    float g[8], p[8];
    if (d.is_bf16) {
      bf16x8 gv = *reinterpret_cast<const bf16x8*>((const short*)d.g + i);
      bf16x8 pv = *reinterpret_cast<const bf16x8*>((const short*)d.p + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        g[j] = bf2f(gv[j]) * grad_scale;
        p[j] = bf2f(pv[j]);
      }
    } else {
      f32x4 g0 = *reinterpret_cast<const f32x4*>((const float*)d.g + i);
      f32x4 g1 = *reinterpret_cast<const f32x4*>((const float*)d.g + i + 4);
      f32x4 p0 = *reinterpret_cast<const f32x4*>((const float*)d.p + i);
      f32x4 p1 = *reinterpret_cast<const f32x4*>((const float*)d.p + i + 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        g[j] = g0[j] * grad_scale;
        g[4 + j] = g1[j] * grad_scale;
        p[j] = p0[j];
        p[4 + j] = p1[j];
      }
    }
    f32x4 m0 = *reinterpret_cast<f32x4*>(d.m + i);
    f32x4 m1 = *reinterpret_cast<f32x4*>(d.m + i + 4);
    f32x4 v0 = *reinterpret_cast<f32x4*>(d.v + i);
    f32x4 v1 = *reinterpret_cast<f32x4*>(d.v + i + 4);
    float m[8] = {m0[0], m0[1], m0[2], m0[3], m1[0], m1[1], m1[2], m1[3]};
    float v[8] = {v0[0], v0[1], v0[2], v0[3], v1[0], v1[1], v1[2], v1[3]};
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      m[j] = m[j] * beta1 + g[j] * (1.0f - beta1);
      v[j] = v[j] * beta2 + g[j] * g[j] * (1.0f - beta2);
      p[j] = p[j] * wd_factor - lr * (m[j] / bc1) / (sqrtf(v[j] / bc2) + eps);
    }
    *reinterpret_cast<f32x4*>(d.m + i) = f32x4{m[0], m[1], m[2], m[3]};
    *reinterpret_cast<f32x4*>(d.m + i + 4) = f32x4{m[4], m[5], m[6], m[7]};
    *reinterpret_cast<f32x4*>(d.v + i) = f32x4{v[0], v[1], v[2], v[3]};
    *reinterpret_cast<f32x4*>(d.v + i + 4) = f32x4{v[4], v[5], v[6], v[7]};
    if (d.is_bf16) {
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) o[j] = f2bf(p[j]);
      *reinterpret_cast<bf16x8*>((short*)d.p + i) = o;
    } else {
      *reinterpret_cast<f32x4*>((float*)d.p + i) =
          f32x4{p[0], p[1], p[2], p[3]};
      *reinterpret_cast<f32x4*>((float*)d.p + i + 4) =
          f32x4{p[4], p[5], p[6], p[7]};
    }
  }
  // scalar tail (final partial 8-group of the tensor's last chunk)
  for (int64_t i = base + n_full + threadIdx.x; i < end; i += ADAM_BLOCK) {
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
