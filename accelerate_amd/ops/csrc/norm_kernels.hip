// Fused LayerNorm / RMSNorm for CDNA4, bf16 in/out with fp32 statistics.
//
// One wave (64 lanes) per row; 16 B/lane vectorized loads (Guideline 13).
// Row data is re-read in the normalize pass — a 1.5 KB row is L1-resident,
// so the second pass costs L1 bandwidth, not HBM. Backward reduces dweight/
// dbias through 64 fp32 partial rows (atomic contention 1/64 of direct
// atomics — Guideline 12) + a small fold kernel.

#include "multi_tensor.h"

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

typedef ushort ushort8 __attribute__((ext_vector_type(8)));  // 16 B = 8 bf16

__device__ __forceinline__ float bf2f(ushort u) {
  union { unsigned int i; float f; } cv;
  cv.i = ((unsigned int)u) << 16;
  return cv.f;
}

__device__ __forceinline__ ushort f2bf(float f) {
  union { float f; unsigned int i; } cv;
  cv.f = f;
  if ((cv.i & 0x7F800000u) == 0x7F800000u) return (ushort)(cv.i >> 16) | (ushort)((cv.i & 0xFFFFu) ? 0x40 : 0);
  cv.i += 0x7FFFu + ((cv.i >> 16) & 1u);
  return (ushort)(cv.i >> 16);
}

__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_down(v, off, 64);
  return __shfl(v, 0, 64);
}

}  // namespace

// ---------------------------------------------------------------------------
// LayerNorm forward: y = (x - mean) * rstd * w + b
// grid = n_rows blocks of 64 threads
// ---------------------------------------------------------------------------

__global__ void layernorm_fwd_bf16(const ushort* __restrict__ x, const ushort* __restrict__ w,
                                   const ushort* __restrict__ b, ushort* __restrict__ y,
                                   float* __restrict__ mean_out, float* __restrict__ rstd_out,
                                   int64_t n_rows, int d, float eps) {
  const int64_t row = blockIdx.x;
  if (row >= n_rows) return;
  const ushort* xr = x + row * d;
  ushort* yr = y + row * d;
  const int lane = threadIdx.x;

  float s = 0.f, sq = 0.f;
  for (int base = lane * 8; base < d; base += 64 * 8) {
    ushort8 av = *reinterpret_cast<const ushort8*>(xr + base);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float v = bf2f(av[k]);
      s += v;
      sq += v * v;
    }
  }
  s = wave_sum(s);
  sq = wave_sum(sq);
  const float mean = s / d;
  const float var = sq / d - mean * mean;
  const float rstd = rsqrtf(var + eps);
  if (lane == 0) {
    mean_out[row] = mean;
    rstd_out[row] = rstd;
  }
  for (int base = lane * 8; base < d; base += 64 * 8) {
    ushort8 av = *reinterpret_cast<const ushort8*>(xr + base);
    ushort8 wv = *reinterpret_cast<const ushort8*>(w + base);
    ushort8 bv = *reinterpret_cast<const ushort8*>(b + base);
    ushort8 ov;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float v = bf2f(av[k]);
      float wk = bf2f(wv[k]);
      float bk = bf2f(bv[k]);
      ushort r = f2bf((v - mean) * rstd * wk + bk);
      ov[k] = r;
    }
    *reinterpret_cast<ushort8*>(yr + base) = ov;
  }
}

// ---------------------------------------------------------------------------
// LayerNorm backward:
//   xhat = (x - mean) * rstd
//   dx = rstd * (dy*w - mean(dy*w) - xhat * mean(dy*w*xhat))
//   dw_partial[row%64] += dy * xhat ; db_partial[row%64] += dy
// ---------------------------------------------------------------------------

// grid-stride over rows: each block accumulates its rows' dw/db in REGISTERS
// (per-lane column slots), then writes ONE non-atomic partial row. No atomics
// anywhere; a fold kernel sums the per-block partials.
// MAX_COLS_PER_LANE bounds d at 64*8*4 = 2048 for LN (BERT 768 fits).
#define LN_MAX_ITERS 4

__global__ void layernorm_bwd_bf16(const ushort* __restrict__ dy, const ushort* __restrict__ x,
                                   const ushort* __restrict__ w,
                                   const float* __restrict__ mean, const float* __restrict__ rstd,
                                   ushort* __restrict__ dx,
                                   float* __restrict__ dw_partial, float* __restrict__ db_partial,
                                   int64_t n_rows, int d) {
  const int lane = threadIdx.x;
  float accw[LN_MAX_ITERS][8];
  float accb[LN_MAX_ITERS][8];
#pragma unroll
  for (int i = 0; i < LN_MAX_ITERS; ++i)
#pragma unroll
    for (int k = 0; k < 8; ++k) { accw[i][k] = 0.f; accb[i][k] = 0.f; }

  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const ushort* dyr = dy + row * d;
    const ushort* xr = x + row * d;
    ushort* dxr = dx + row * d;
    const float mu = mean[row], rs = rstd[row];

    float s1 = 0.f, s2 = 0.f;
    for (int base = lane * 8; base < d; base += 64 * 8) {
      ushort8 gv = *reinterpret_cast<const ushort8*>(dyr + base);
      ushort8 xv8 = *reinterpret_cast<const ushort8*>(xr + base);
      ushort8 wv = *reinterpret_cast<const ushort8*>(w + base);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = bf2f(gv[k]);
        float xv = bf2f(xv8[k]);
        float wk = bf2f(wv[k]);
        float xhat = (xv - mu) * rs;
        float gw = g * wk;
        s1 += gw;
        s2 += gw * xhat;
      }
    }
    s1 = wave_sum(s1) / d;
    s2 = wave_sum(s2) / d;

    int it = 0;
    for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
      ushort8 gv = *reinterpret_cast<const ushort8*>(dyr + base);
      ushort8 xv8 = *reinterpret_cast<const ushort8*>(xr + base);
      ushort8 wv = *reinterpret_cast<const ushort8*>(w + base);
      ushort8 ov;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = bf2f(gv[k]);
        float xv = bf2f(xv8[k]);
        float wk = bf2f(wv[k]);
        float xhat = (xv - mu) * rs;
        float dxv = rs * (g * wk - s1 - xhat * s2);
        ushort r = f2bf(dxv);
        ov[k] = r;
        accw[it][k] += g * xhat;
        accb[it][k] += g;
      }
      *reinterpret_cast<ushort8*>(dxr + base) = ov;
    }
  }

  // one partial row per block, non-atomic
  float* dwp = dw_partial + blockIdx.x * (int64_t)d;
  float* dbp = db_partial + blockIdx.x * (int64_t)d;
  int it = 0;
  for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      dwp[base + k] = accw[it][k];
      dbp[base + k] = accb[it][k];
    }
  }
}

// fold `n_partials` partial rows into bf16 dw/db
__global__ void norm_fold_partials(const float* __restrict__ dw_partial, const float* __restrict__ db_partial,
                                   ushort* __restrict__ dw, ushort* __restrict__ db, int d, int n_partials) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= d) return;
  float sw = 0.f, sb = 0.f;
  for (int r = 0; r < n_partials; ++r) {
    sw += dw_partial[(int64_t)r * d + col];
    if (db_partial != nullptr) sb += db_partial[(int64_t)r * d + col];
  }
  dw[col] = f2bf(sw);
  if (db != nullptr) db[col] = f2bf(sb);
}

// ---------------------------------------------------------------------------
// RMSNorm: y = x * rstd * w ; rstd = rsqrt(mean(x²)+eps)
//   dx = rstd*dy*w - x * (sum(dy*w*x) * rstd³ / d)
//   dw_partial += dy * x * rstd
// ---------------------------------------------------------------------------

__global__ void rmsnorm_fwd_bf16(const ushort* __restrict__ x, const ushort* __restrict__ w,
                                 ushort* __restrict__ y, float* __restrict__ rstd_out,
                                 int64_t n_rows, int d, float eps) {
  const int64_t row = blockIdx.x;
  if (row >= n_rows) return;
  const ushort* xr = x + row * d;
  ushort* yr = y + row * d;
  const int lane = threadIdx.x;
  float sq = 0.f;
  for (int base = lane * 8; base < d; base += 64 * 8) {
    ushort8 av = *reinterpret_cast<const ushort8*>(xr + base);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float v = bf2f(av[k]);
      sq += v * v;
    }
  }
  sq = wave_sum(sq);
  const float rstd = rsqrtf(sq / d + eps);
  if (lane == 0) rstd_out[row] = rstd;
  for (int base = lane * 8; base < d; base += 64 * 8) {
    ushort8 av = *reinterpret_cast<const ushort8*>(xr + base);
    ushort8 wv = *reinterpret_cast<const ushort8*>(w + base);
    ushort8 ov;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float v = bf2f(av[k]);
      float wk = bf2f(wv[k]);
      ushort r = f2bf(v * rstd * wk);
      ov[k] = r;
    }
    *reinterpret_cast<ushort8*>(yr + base) = ov;
  }
}

// grid-stride rows + register dw accumulation (see layernorm_bwd).
// RMS_MAX_ITERS=16 bounds d at 64*8*16 = 8192 (Llama-70B hidden).
#define RMS_MAX_ITERS 16

__global__ void rmsnorm_bwd_bf16(const ushort* __restrict__ dy, const ushort* __restrict__ x,
                                 const ushort* __restrict__ w, const float* __restrict__ rstd,
                                 ushort* __restrict__ dx, float* __restrict__ dw_partial,
                                 int64_t n_rows, int d) {
  const int lane = threadIdx.x;
  const int n_iters = (d + 64 * 8 - 1) / (64 * 8);
  float accw[RMS_MAX_ITERS][8];
  for (int i = 0; i < n_iters; ++i)
#pragma unroll
    for (int k = 0; k < 8; ++k) accw[i][k] = 0.f;

  for (int64_t row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const ushort* dyr = dy + row * d;
    const ushort* xr = x + row * d;
    ushort* dxr = dx + row * d;
    const float rs = rstd[row];

    float s = 0.f;
    for (int base = lane * 8; base < d; base += 64 * 8) {
      ushort8 gv = *reinterpret_cast<const ushort8*>(dyr + base);
      ushort8 xv8 = *reinterpret_cast<const ushort8*>(xr + base);
      ushort8 wv = *reinterpret_cast<const ushort8*>(w + base);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = bf2f(gv[k]);
        float xv = bf2f(xv8[k]);
        float wk = bf2f(wv[k]);
        s += g * wk * xv;
      }
    }
    s = wave_sum(s);
    const float c = s * rs * rs * rs / d;

    int it = 0;
    for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
      ushort8 gv = *reinterpret_cast<const ushort8*>(dyr + base);
      ushort8 xv8 = *reinterpret_cast<const ushort8*>(xr + base);
      ushort8 wv = *reinterpret_cast<const ushort8*>(w + base);
      ushort8 ov;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = bf2f(gv[k]);
        float xv = bf2f(xv8[k]);
        float wk = bf2f(wv[k]);
        float dxv = rs * g * wk - xv * c;
        ushort r = f2bf(dxv);
        ov[k] = r;
        accw[it][k] += g * xv * rs;
      }
      *reinterpret_cast<ushort8*>(dxr + base) = ov;
    }
  }

  float* dwp = dw_partial + blockIdx.x * (int64_t)d;
  int it = 0;
  for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
#pragma unroll
    for (int k = 0; k < 8; ++k) dwp[base + k] = accw[it][k];
  }
}
