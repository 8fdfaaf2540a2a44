// LayerNorm forward/backward for bf16 rows (gfx950).
//
// Memory-bound: target HBM ceiling via bf16x8 vector loads (guide G13),
// fp32 accumulation, one workgroup per row (fwd/dx), striped column-partial
// reduction for dw/db (deterministic — no atomics, so TP replicas stay
// bitwise identical).
//
// Covers the reference's LayerNorm fwd/bwd fusion slot (SURVEY.md §2.3
// kernel table: "LayerNorm fwd/bwd | elementwise+reduce fusion | [B·S, H]").
#include "common.h"

#define LN_BLOCK 256

// ---------------------------------------------------------------- forward
// x [N, H] bf16, w/b [H] bf16 -> y [N, H] bf16, mean/rstd [N] fp32
__global__ void layer_norm_fwd_kernel(const short* __restrict__ x,
                                      const short* __restrict__ w,
                                      const short* __restrict__ b,
                                      short* __restrict__ y,
                                      float* __restrict__ mean_out,
                                      float* __restrict__ rstd_out,
                                      int H, float eps) {
  const int row = blockIdx.x;
  const short* xr = x + (int64_t)row * H;
  short* yr = y + (int64_t)row * H;
  __shared__ float red[LN_BLOCK / 64];

  // pass 1: sum & sumsq (vectorized)
  float s = 0.f, ss = 0.f;
  for (int i = threadIdx.x * 8; i < H; i += LN_BLOCK * 8) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float f = bf2f(v[j]);
      s += f;
      ss += f * f;
    }
  }
  s = block_reduce_sum(s, red);
  ss = block_reduce_sum(ss, red);
  const float mean = s / H;
  const float var = ss / H - mean * mean;
  const float rstd = rsqrtf(var + eps);
  if (threadIdx.x == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }

  // pass 2: normalize (x re-read hits L2/L3)
  for (int i = threadIdx.x * 8; i < H; i += LN_BLOCK * 8) {
    bf16x8 v = *reinterpret_cast<const bf16x8*>(xr + i);
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + i);
    bf16x8 bv = *reinterpret_cast<const bf16x8*>(b + i);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xhat = (bf2f(v[j]) - mean) * rstd;
      o[j] = f2bf(xhat * bf2f(wv[j]) + bf2f(bv[j]));
    }
    *reinterpret_cast<bf16x8*>(yr + i) = o;
  }
}

// ---------------------------------------------------------------- backward dx
// dx = (wdy - mean(wdy) - xhat * mean(wdy*xhat)) * rstd
__global__ void layer_norm_bwd_dx_kernel(const short* __restrict__ dy,
                                         const short* __restrict__ x,
                                         const short* __restrict__ w,
                                         const float* __restrict__ mean_in,
                                         const float* __restrict__ rstd_in,
                                         short* __restrict__ dx, int H) {
  const int row = blockIdx.x;
  const short* dyr = dy + (int64_t)row * H;
  const short* xr = x + (int64_t)row * H;
  short* dxr = dx + (int64_t)row * H;
  const float mean = mean_in[row], rstd = rstd_in[row];
  __shared__ float red[LN_BLOCK / 64];

  float c1 = 0.f, c2 = 0.f;
  for (int i = threadIdx.x * 8; i < H; i += LN_BLOCK * 8) {
    bf16x8 dv = *reinterpret_cast<const bf16x8*>(dyr + i);
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(xr + i);
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float wdy = bf2f(dv[j]) * bf2f(wv[j]);
      float xhat = (bf2f(xv[j]) - mean) * rstd;
      c1 += wdy;
      c2 += wdy * xhat;
    }
  }
  c1 = block_reduce_sum(c1, red) / H;
  c2 = block_reduce_sum(c2, red) / H;

  for (int i = threadIdx.x * 8; i < H; i += LN_BLOCK * 8) {
    bf16x8 dv = *reinterpret_cast<const bf16x8*>(dyr + i);
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(xr + i);
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + i);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float wdy = bf2f(dv[j]) * bf2f(wv[j]);
      float xhat = (bf2f(xv[j]) - mean) * rstd;
      o[j] = f2bf((wdy - c1 - xhat * c2) * rstd);
    }
    *reinterpret_cast<bf16x8*>(dxr + i) = o;
  }
}

// ------------------------------------------------------------- backward dw/db
// Striped partials: grid (H/CHUNK, P); block (r, p) accumulates rows
// r*? — rows p, p+P, p+2P... over column chunk. Output partial [P, H] fp32;
// final sum over P done by a tiny second kernel.
#define LNB_COLS 256  // columns per block (1 per thread)

__global__ void layer_norm_bwd_dwdb_partial(const short* __restrict__ dy,
                                            const short* __restrict__ x,
                                            const float* __restrict__ mean_in,
                                            const float* __restrict__ rstd_in,
                                            float* __restrict__ dw_part,
                                            float* __restrict__ db_part,
                                            int N, int H) {
  // 4 columns per thread (bf16x4 loads — the 1-col scalar version was
  // issue-bound well off the HBM roofline, cf. the dbias partial)
  const int col4 = (blockIdx.x * LNB_COLS + threadIdx.x) * 4;
  const int p = blockIdx.y;  // stripe index
  const int P = gridDim.y;
  // contiguous row range per stripe (DRAM page locality)
  const int64_t r0 = (int64_t)p * N / P, r1 = (int64_t)(p + 1) * N / P;
  if (col4 + 4 <= H) {
    float dw[4] = {0.f, 0.f, 0.f, 0.f}, db[4] = {0.f, 0.f, 0.f, 0.f};
    for (int64_t row = r0; row < r1; ++row) {
      float mean = mean_in[row], rstd = rstd_in[row];
      bf16x4 dv = *reinterpret_cast<const bf16x4*>(dy + row * H + col4);
      bf16x4 xv = *reinterpret_cast<const bf16x4*>(x + row * H + col4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float d = bf2f(dv[j]);
        dw[j] += d * (bf2f(xv[j]) - mean) * rstd;
        db[j] += d;
      }
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      dw_part[(int64_t)p * H + col4 + j] = dw[j];
      db_part[(int64_t)p * H + col4 + j] = db[j];
    }
  } else if (col4 < H) {
    for (int c = col4; c < H; ++c) {
      float dw = 0.f, db = 0.f;
      for (int64_t row = r0; row < r1; ++row) {
        float mean = mean_in[row], rstd = rstd_in[row];
        float d = bf2f(dy[row * H + c]);
        dw += d * (bf2f(x[row * H + c]) - mean) * rstd;
        db += d;
      }
      dw_part[(int64_t)p * H + c] = dw;
      db_part[(int64_t)p * H + c] = db;
    }
  }
}

__global__ void column_sum_kernel(const float* __restrict__ part,
                                  float* __restrict__ out, int P, int H) {
  int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= H) return;
  float s = 0.f;
  for (int p = 0; p < P; ++p) s += part[(int64_t)p * H + col];
  out[col] = s;
}

// ---------------------------------------------------------------- launchers
extern "C" {

hipError_t launch_layer_norm_fwd(const void* x, const void* w, const void* b,
                                 void* y, float* mean, float* rstd, int64_t N,
                                 int64_t H, float eps, hipStream_t stream) {
  layer_norm_fwd_kernel<<<dim3((uint32_t)N), dim3(LN_BLOCK), 0, stream>>>(
      (const short*)x, (const short*)w, (const short*)b, (short*)y, mean,
      rstd, (int)H, eps);
  return hipGetLastError();
}

hipError_t launch_layer_norm_bwd_dx(const void* dy, const void* x,
                                    const void* w, const float* mean,
                                    const float* rstd, void* dx, int64_t N,
                                    int64_t H, hipStream_t stream) {
  layer_norm_bwd_dx_kernel<<<dim3((uint32_t)N), dim3(LN_BLOCK), 0, stream>>>(
      (const short*)dy, (const short*)x, (const short*)w, mean, rstd,
      (short*)dx, (int)H);
  return hipGetLastError();
}

hipError_t launch_layer_norm_bwd_dwdb(const void* dy, const void* x,
                                      const float* mean, const float* rstd,
                                      float* dw_part, float* db_part,
                                      float* /*dw*/, float* /*db*/, int64_t N,
                                      int64_t H, int P, hipStream_t stream) {
  dim3 grid((uint32_t)ceil_div(H, LNB_COLS * 4), P);
  layer_norm_bwd_dwdb_partial<<<grid, dim3(LNB_COLS), 0, stream>>>(
      (const short*)dy, (const short*)x, mean, rstd, dw_part, db_part,
      (int)N, (int)H);
  // final P-row reduction handled by the caller (at::sum — parallel
  // reduce kernel beats the serial per-column loop)
  return hipGetLastError();
}

}  // extern "C"

// ------------------------------------------------------------------
// Fused residual-add + LayerNorm: h = a + b; y = LN(h) * w + bias.
// Saves the separate elementwise-add pass over the residual stream
// (the reference gets this from XLA fusion).  b may be null (plain LN
// with h written = a, used at the embedding boundary).
__global__ void add_layer_norm_fwd_kernel(const short* __restrict__ a,
                                          const short* __restrict__ b,
                                          const short* __restrict__ w,
                                          const short* __restrict__ bias,
                                          short* __restrict__ h,
                                          short* __restrict__ y,
                                          float* __restrict__ mean_out,
                                          float* __restrict__ rstd_out,
                                          int H, float eps) {
  const int row = blockIdx.x;
  const short* ar = a + (int64_t)row * H;
  const short* br = (b == nullptr) ? nullptr : b + (int64_t)row * H;
  short* hr = h + (int64_t)row * H;
  short* yr = y + (int64_t)row * H;
  __shared__ float red[LN_BLOCK / 64];

  float s = 0.f, ss = 0.f;
  for (int i = threadIdx.x * 8; i < H; i += LN_BLOCK * 8) {
    bf16x8 av = *reinterpret_cast<const bf16x8*>(ar + i);
    bf16x8 hv;
    if (br != nullptr) {
      bf16x8 bv = *reinterpret_cast<const bf16x8*>(br + i);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(av[j]) + bf2f(bv[j]);
        hv[j] = f2bf(f);
        s += f;
        ss += f * f;
      }
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(av[j]);
        hv[j] = av[j];
        s += f;
        ss += f * f;
      }
    }
    *reinterpret_cast<bf16x8*>(hr + i) = hv;
  }
  s = block_reduce_sum(s, red);
  ss = block_reduce_sum(ss, red);
  const float mean = s / H;
  const float rstd = rsqrtf(ss / H - mean * mean + eps);
  if (threadIdx.x == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  for (int i = threadIdx.x * 8; i < H; i += LN_BLOCK * 8) {
    bf16x8 hv = *reinterpret_cast<const bf16x8*>(hr + i);
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + i);
    bf16x8 bv2 = *reinterpret_cast<const bf16x8*>(bias + i);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float xhat = (bf2f(hv[j]) - mean) * rstd;
      o[j] = f2bf(xhat * bf2f(wv[j]) + bf2f(bv2[j]));
    }
    *reinterpret_cast<bf16x8*>(yr + i) = o;
  }
}

// backward dx with an extra residual-gradient term:
// g = LN_bwd_dx(dy) + dh   (dh may be null)
__global__ void add_layer_norm_bwd_dx_kernel(
    const short* __restrict__ dy, const short* __restrict__ dh,
    const short* __restrict__ x, const short* __restrict__ w,
    const float* __restrict__ mean_in, const float* __restrict__ rstd_in,
    short* __restrict__ dx, int H) {
  const int row = blockIdx.x;
  const short* dyr = dy + (int64_t)row * H;
  const short* dhr = (dh == nullptr) ? nullptr : dh + (int64_t)row * H;
  const short* xr = x + (int64_t)row * H;
  short* dxr = dx + (int64_t)row * H;
  const float mean = mean_in[row], rstd = rstd_in[row];
  __shared__ float red[LN_BLOCK / 64];

  // two passes over the row (L2-resident for these row sizes; a
  // register-cached single-read variant measured NEUTRAL-to-worse —
  // the extra VGPRs bought nothing).  dh (residual grad) read is
  // vectorized.
  float c1 = 0.f, c2 = 0.f;
  for (int i = threadIdx.x * 8; i < H; i += LN_BLOCK * 8) {
    bf16x8 dv = *reinterpret_cast<const bf16x8*>(dyr + i);
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(xr + i);
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + i);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float wdy = bf2f(dv[j]) * bf2f(wv[j]);
      float xhat = (bf2f(xv[j]) - mean) * rstd;
      c1 += wdy;
      c2 += wdy * xhat;
    }
  }
  c1 = block_reduce_sum(c1, red) / H;
  c2 = block_reduce_sum(c2, red) / H;

  for (int i = threadIdx.x * 8; i < H; i += LN_BLOCK * 8) {
    bf16x8 dv = *reinterpret_cast<const bf16x8*>(dyr + i);
    bf16x8 xv = *reinterpret_cast<const bf16x8*>(xr + i);
    bf16x8 wv = *reinterpret_cast<const bf16x8*>(w + i);
    bf16x8 hv;
    if (dhr != nullptr)
      hv = *reinterpret_cast<const bf16x8*>(dhr + i);
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float wdy = bf2f(dv[j]) * bf2f(wv[j]);
      float xhat = (bf2f(xv[j]) - mean) * rstd;
      float g = (wdy - c1 - xhat * c2) * rstd;
      if (dhr != nullptr) g += bf2f(hv[j]);
      o[j] = f2bf(g);
    }
    *reinterpret_cast<bf16x8*>(dxr + i) = o;
  }
}

extern "C" {

hipError_t launch_add_layer_norm_fwd(const void* a, const void* b,
                                     const void* w, const void* bias,
                                     void* h, void* y, float* mean,
                                     float* rstd, int64_t N, int64_t H,
                                     float eps, hipStream_t stream) {
  add_layer_norm_fwd_kernel<<<dim3((uint32_t)N), dim3(LN_BLOCK), 0,
                              stream>>>(
      (const short*)a, (const short*)b, (const short*)w, (const short*)bias,
      (short*)h, (short*)y, mean, rstd, (int)H, eps);
  return hipGetLastError();
}

hipError_t launch_add_layer_norm_bwd_dx(const void* dy, const void* dh,
                                        const void* x, const void* w,
                                        const float* mean, const float* rstd,
                                        void* dx, int64_t N, int64_t H,
                                        hipStream_t stream) {
  add_layer_norm_bwd_dx_kernel<<<dim3((uint32_t)N), dim3(LN_BLOCK), 0,
                                 stream>>>(
      (const short*)dy, (const short*)dh, (const short*)x, (const short*)w,
      mean, rstd, (short*)dx, (int)H);
  return hipGetLastError();
}

}  // extern "C"
