// Fused LayerNorm / RMSNorm for CDNA4, bf16 in/out with fp32 statistics.
//
// Forward: one wave (64 lanes) per row; 16 B/lane vectorized loads
// (Guideline 13); the 1.5 KB row is L1-resident for the normalize pass.
// Backward: 256-thread blocks run 4 rows wave-per-row (the one-wave-per-
// block shape left half the 1024 SIMDs idle); small d keeps dy/x in
// registers for a single HBM pass; dweight/dbias accumulate in per-lane
// register slots -> one non-atomic fp32 partial row per wave -> fold
// kernel. d > 2048 (RMS) splits dx from a column-tiled dw kernel.

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

// 256-thread blocks, one WAVE per row (4 rows in flight per block): the
// old one-wave-per-block shape put only 512 waves on a 1024-SIMD chip.
// Row data (dy, x) loaded once into registers when d <= ITERS*512 fits
// (KEEP=true), so stats + dx are a single HBM pass; dweight/dbias
// accumulate in per-lane register slots and each wave writes ONE
// non-atomic partial row; the host folds partials with torch sum(0).
template <int ITERS, bool KEEP>
__global__ __launch_bounds__(256) void layernorm_bwd_t(
    const ushort* __restrict__ dy, const ushort* __restrict__ x,
    const ushort* __restrict__ w,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    ushort* __restrict__ dx,
    float* __restrict__ dw_partial, float* __restrict__ db_partial,
    int64_t n_rows, int d) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  float accw[ITERS][8];
  float accb[ITERS][8];
#pragma unroll
  for (int i = 0; i < ITERS; ++i)
#pragma unroll
    for (int k = 0; k < 8; ++k) { accw[i][k] = 0.f; accb[i][k] = 0.f; }

  float xv[KEEP ? ITERS : 1][8];
  float gv[KEEP ? ITERS : 1][8];

  for (int64_t row = (int64_t)blockIdx.x * 4 + wave; row < n_rows; row += (int64_t)gridDim.x * 4) {
    const ushort* dyr = dy + row * d;
    const ushort* xr = x + row * d;
    ushort* dxr = dx + row * d;
    const float mu = mean[row], rs = rstd[row];

    // pass 1: row stats. KEEP stashes RAW dy and x in registers; w is
    // re-read in pass 2 (shared across all rows — L1-resident).
    float s1 = 0.f, s2 = 0.f;
    int it = 0;
    for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
      ushort8 g8 = *reinterpret_cast<const ushort8*>(dyr + base);
      ushort8 x8 = *reinterpret_cast<const ushort8*>(xr + base);
      ushort8 w8 = *reinterpret_cast<const ushort8*>(w + base);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = bf2f(g8[k]);
        float v = bf2f(x8[k]);
        if (KEEP) { gv[KEEP ? it : 0][k] = g; xv[KEEP ? it : 0][k] = v; }
        float gw = g * bf2f(w8[k]);
        s1 += gw;
        s2 += gw * (v - mu) * rs;
      }
    }
    s1 = wave_sum(s1) / d;
    s2 = wave_sum(s2) / d;

    it = 0;
    for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
      float gk[8], vk[8];
      if (!KEEP) {
        ushort8 g8 = *reinterpret_cast<const ushort8*>(dyr + base);
        ushort8 x8 = *reinterpret_cast<const ushort8*>(xr + base);
#pragma unroll
        for (int k = 0; k < 8; ++k) { gk[k] = bf2f(g8[k]); vk[k] = bf2f(x8[k]); }
      } else {
#pragma unroll
        for (int k = 0; k < 8; ++k) { gk[k] = gv[KEEP ? it : 0][k]; vk[k] = xv[KEEP ? it : 0][k]; }
      }
      ushort8 w8 = *reinterpret_cast<const ushort8*>(w + base);
      ushort8 ov;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float xhat = (vk[k] - mu) * rs;
        ov[k] = f2bf(rs * (gk[k] * bf2f(w8[k]) - s1 - xhat * s2));
        accw[it][k] += gk[k] * xhat;
        accb[it][k] += gk[k];
      }
      *reinterpret_cast<ushort8*>(dxr + base) = ov;
    }
  }

  // one partial row per WAVE, non-atomic; dw and db share one [P][2d]
  // buffer so the host folds both with a single sum(0) launch
  const int64_t pidx = (int64_t)blockIdx.x * 4 + wave;
  float* dwp = dw_partial + pidx * (2 * (int64_t)d);
  int it = 0;
  for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      dwp[base + k] = accw[it][k];
      dwp[d + base + k] = accb[it][k];
    }
  }
}

// host-side dispatch over the templated iteration counts (d <= ITERS*512)
extern "C" hipError_t launch_layernorm_bwd(const void* dy, const void* x, const void* w,
                                           const void* mean, const void* rstd, void* dx,
                                           void* dw_partial, void* db_partial,
                                           long long n_rows, int d, int n_blocks,
                                           hipStream_t stream) {
  dim3 g(n_blocks), b(256);
#define LN_CASE(N, KEEP) \
  hipLaunchKernelGGL((layernorm_bwd_t<N, KEEP>), g, b, 0, stream, \
                     (const ushort*)dy, (const ushort*)x, (const ushort*)w, \
                     (const float*)mean, (const float*)rstd, (ushort*)dx, \
                     (float*)dw_partial, (float*)db_partial, (int64_t)n_rows, d)
  if (d <= 512) LN_CASE(1, true);
  else if (d <= 1024) LN_CASE(2, true);
  else if (d <= 2048) LN_CASE(4, false);
  else return hipErrorInvalidValue;
#undef LN_CASE
  return hipGetLastError();
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

// 256-thread blocks, one WAVE per row (see layernorm_bwd_t). KEEP=true
// holds dy/x in registers across the stat + dx passes (d <= ITERS*512);
// larger d (Llama 4096/8192) re-reads the L1/L2-resident row.
// Decode-shape forward (n_rows < 4): all 256 threads cooperate on ONE
// row (the wave-per-row kernel would leave 3 of 4 waves idle and stream a
// 16 KB row through a single wave — measured 17.5 us/call on 70B decode).
__global__ __launch_bounds__(256) void rmsnorm_fwd_wide(
    const ushort* __restrict__ x, const ushort* __restrict__ w,
    ushort* __restrict__ y, float* __restrict__ rstd_out,
    int64_t n_rows, int d, float eps) {
  __shared__ float red[4];
  const int64_t row = blockIdx.x;
  if (row >= n_rows) return;
  const ushort* xr = x + row * d;
  ushort* yr = y + row * d;
  const int tid = threadIdx.x;
  float sq = 0.f;
  for (int base = tid * 8; base < d; base += 256 * 8) {
    ushort8 av = *reinterpret_cast<const ushort8*>(xr + base);
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float v = bf2f(av[k]);
      sq += v * v;
    }
  }
  sq = wave_sum(sq);
  if ((tid & 63) == 0) red[tid >> 6] = sq;
  __syncthreads();
  const float rstd = rsqrtf((red[0] + red[1] + red[2] + red[3]) / d + eps);
  if (tid == 0) rstd_out[row] = rstd;
  for (int base = tid * 8; base < d; base += 256 * 8) {
    ushort8 av = *reinterpret_cast<const ushort8*>(xr + base);
    ushort8 wv = *reinterpret_cast<const ushort8*>(w + base);
    ushort8 ov;
#pragma unroll
    for (int k = 0; k < 8; ++k) ov[k] = f2bf(bf2f(av[k]) * rstd * bf2f(wv[k]));
    *reinterpret_cast<ushort8*>(yr + base) = ov;
  }
}

template <int ITERS, bool KEEP>
__global__ __launch_bounds__(256) void rmsnorm_bwd_t(
    const ushort* __restrict__ dy, const ushort* __restrict__ x,
    const ushort* __restrict__ w, const float* __restrict__ rstd,
    ushort* __restrict__ dx, float* __restrict__ dw_partial,
    int64_t n_rows, int d) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  float accw[ITERS][8];
#pragma unroll
  for (int i = 0; i < ITERS; ++i)
#pragma unroll
    for (int k = 0; k < 8; ++k) accw[i][k] = 0.f;

  float xv[KEEP ? ITERS : 1][8];
  float gv[KEEP ? ITERS : 1][8];

  for (int64_t row = (int64_t)blockIdx.x * 4 + wave; row < n_rows; row += (int64_t)gridDim.x * 4) {
    const ushort* dyr = dy + row * d;
    const ushort* xr = x + row * d;
    ushort* dxr = dx + row * d;
    const float rs = rstd[row];

    float sum = 0.f;
    int it = 0;
    for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
      ushort8 g8 = *reinterpret_cast<const ushort8*>(dyr + base);
      ushort8 x8 = *reinterpret_cast<const ushort8*>(xr + base);
      ushort8 w8 = *reinterpret_cast<const ushort8*>(w + base);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = bf2f(g8[k]);
        float v = bf2f(x8[k]);
        if (KEEP) { gv[KEEP ? it : 0][k] = g; xv[KEEP ? it : 0][k] = v; }
        sum += g * bf2f(w8[k]) * v;
      }
    }
    sum = wave_sum(sum);
    const float c = sum * rs * rs * rs / d;

    it = 0;
    for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
      float gk[8], vk[8];
      if (!KEEP) {
        ushort8 g8 = *reinterpret_cast<const ushort8*>(dyr + base);
        ushort8 x8 = *reinterpret_cast<const ushort8*>(xr + base);
#pragma unroll
        for (int k = 0; k < 8; ++k) { gk[k] = bf2f(g8[k]); vk[k] = bf2f(x8[k]); }
      } else {
#pragma unroll
        for (int k = 0; k < 8; ++k) { gk[k] = gv[KEEP ? it : 0][k]; vk[k] = xv[KEEP ? it : 0][k]; }
      }
      ushort8 w8 = *reinterpret_cast<const ushort8*>(w + base);
      ushort8 ov;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        ov[k] = f2bf(rs * gk[k] * bf2f(w8[k]) - vk[k] * c);
        accw[it][k] += gk[k] * vk[k] * rs;
      }
      *reinterpret_cast<ushort8*>(dxr + base) = ov;
    }
  }

  float* dwp = dw_partial + ((int64_t)blockIdx.x * 4 + wave) * d;
  int it = 0;
  for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
#pragma unroll
    for (int k = 0; k < 8; ++k) dwp[base + k] = accw[it][k];
  }
}

// Large-d (> 2048) RMS backward: the fused kernel's per-lane dw slots no
// longer fit in registers (ITERS 8/16 spilled to scratch), so split into a
// lean dx-only kernel (stats + dx, no accumulators — occupancy 8) and a
// column-tiled dw kernel. dweight needs NO row statistics beyond the saved
// mean/rstd, so the dw grid is 2D: blockIdx.y picks a 512-column tile
// (one ushort8 per lane), blockIdx.x grid-strides rows wave-per-row; each
// (row-block, tile) writes its 512-column slice of a full-width partial.
__global__ __launch_bounds__(256) void rmsnorm_bwd_dx(
    const ushort* __restrict__ dy, const ushort* __restrict__ x,
    const ushort* __restrict__ w, const float* __restrict__ rstd,
    ushort* __restrict__ dx, int64_t n_rows, int d) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  for (int64_t row = (int64_t)blockIdx.x * 4 + wave; row < n_rows; row += (int64_t)gridDim.x * 4) {
    const ushort* dyr = dy + row * d;
    const ushort* xr = x + row * d;
    ushort* dxr = dx + row * d;
    const float rs = rstd[row];
    float sum = 0.f;
    for (int base = lane * 8; base < d; base += 64 * 8) {
      ushort8 g8 = *reinterpret_cast<const ushort8*>(dyr + base);
      ushort8 x8 = *reinterpret_cast<const ushort8*>(xr + base);
      ushort8 w8 = *reinterpret_cast<const ushort8*>(w + base);
#pragma unroll
      for (int k = 0; k < 8; ++k) sum += bf2f(g8[k]) * bf2f(w8[k]) * bf2f(x8[k]);
    }
    sum = wave_sum(sum);
    const float c = sum * rs * rs * rs / d;
    for (int base = lane * 8; base < d; base += 64 * 8) {
      ushort8 g8 = *reinterpret_cast<const ushort8*>(dyr + base);
      ushort8 x8 = *reinterpret_cast<const ushort8*>(xr + base);
      ushort8 w8 = *reinterpret_cast<const ushort8*>(w + base);
      ushort8 ov;
#pragma unroll
      for (int k = 0; k < 8; ++k)
        ov[k] = f2bf(rs * bf2f(g8[k]) * bf2f(w8[k]) - bf2f(x8[k]) * c);
      *reinterpret_cast<ushort8*>(dxr + base) = ov;
    }
  }
}

// dw (and optionally db) over a 512-column tile; mean == nullptr => RMS
// semantics (xhat = x * rstd), else LayerNorm (xhat = (x - mean) * rstd).
__global__ __launch_bounds__(256) void norm_dw_tile(
    const ushort* __restrict__ dy, const ushort* __restrict__ x,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    float* __restrict__ dw_partial, float* __restrict__ db_partial,
    int64_t n_rows, int d) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int base = blockIdx.y * 512 + lane * 8;
  float accw[8], accb[8];
#pragma unroll
  for (int k = 0; k < 8; ++k) { accw[k] = 0.f; accb[k] = 0.f; }
  if (base < d) {
    for (int64_t row = (int64_t)blockIdx.x * 4 + wave; row < n_rows; row += (int64_t)gridDim.x * 4) {
      ushort8 g8 = *reinterpret_cast<const ushort8*>(dy + row * d + base);
      ushort8 x8 = *reinterpret_cast<const ushort8*>(x + row * d + base);
      const float mu = mean ? mean[row] : 0.f;
      const float rs = rstd[row];
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = bf2f(g8[k]);
        accw[k] += g * (bf2f(x8[k]) - mu) * rs;
        accb[k] += g;
      }
    }
    float* dwp = dw_partial + ((int64_t)blockIdx.x * 4 + wave) * d + base;
#pragma unroll
    for (int k = 0; k < 8; ++k) dwp[k] = accw[k];
    if (db_partial != nullptr) {
      float* dbp = db_partial + ((int64_t)blockIdx.x * 4 + wave) * d + base;
#pragma unroll
      for (int k = 0; k < 8; ++k) dbp[k] = accb[k];
    }
  }
}

extern "C" hipError_t launch_rmsnorm_bwd(const void* dy, const void* x, const void* w,
                                         const void* rstd, void* dx, void* dw_partial,
                                         long long n_rows, int d, int n_blocks,
                                         hipStream_t stream) {
  dim3 g(n_blocks), b(256);
#define RMS_CASE(N, KEEP) \
  hipLaunchKernelGGL((rmsnorm_bwd_t<N, KEEP>), g, b, 0, stream, \
                     (const ushort*)dy, (const ushort*)x, (const ushort*)w, \
                     (const float*)rstd, (ushort*)dx, (float*)dw_partial, (int64_t)n_rows, d)
  if (d <= 512) RMS_CASE(1, true);
  else if (d <= 1024) RMS_CASE(2, true);
  else if (d <= 2048) RMS_CASE(4, false);
  else if (d <= 8192) {
    hipLaunchKernelGGL(rmsnorm_bwd_dx, g, b, 0, stream, (const ushort*)dy, (const ushort*)x,
                       (const ushort*)w, (const float*)rstd, (ushort*)dx, (int64_t)n_rows, d);
    dim3 g2(n_blocks, (d + 511) / 512);
    hipLaunchKernelGGL(norm_dw_tile, g2, b, 0, stream, (const ushort*)dy, (const ushort*)x,
                       (const float*)nullptr, (const float*)rstd, (float*)dw_partial,
                       (float*)nullptr, (int64_t)n_rows, d);
  } else return hipErrorInvalidValue;
#undef RMS_CASE
  return hipGetLastError();
}

// ---------------------------------------------------------------------------
// Fused residual junction: y = LayerNorm(x + dropout(z)) — BERT runs this
// TWICE per layer; unfused it costs a dropout kernel, an add, the LN, and
// their backward mirrors plus intermediate round-trips. One pass here:
// dropout mask is generated in-kernel from the torch philox state
// (graph-capture-safe: ATen's PhiloxCudaState offset is read via pointer
// under capture), the summed input s is written once for the backward,
// and the mask is saved as u8.
// ---------------------------------------------------------------------------

#include <ATen/hip/PhiloxUtils.cuh>

namespace {
__device__ __forceinline__ unsigned mix_rand24(uint64_t seed, uint64_t idx) {
  uint64_t zz = seed + idx * 0x9E3779B97F4A7C15ull;
  zz = (zz ^ (zz >> 30)) * 0xBF58476D1CE4E5B9ull;
  zz = (zz ^ (zz >> 27)) * 0x94D049BB133111EBull;
  return (unsigned)(zz >> 40);  // 24 uniform bits
}
}  // namespace

template <int ITERS, bool KEEP>
__global__ __launch_bounds__(256) void dropout_add_ln_fwd_t(
    const ushort* __restrict__ x, const ushort* __restrict__ z,
    const ushort* __restrict__ w, const ushort* __restrict__ b,
    ushort* __restrict__ y, ushort* __restrict__ s_out,
    unsigned char* __restrict__ mask_out,
    float* __restrict__ mean_out, float* __restrict__ rstd_out,
    int64_t n_rows, int d, float eps, float keep_p, float inv_keep,
    at::PhiloxCudaState rng) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const auto seeds = at::cuda::philox::unpack(rng);
  const uint64_t seed = std::get<0>(seeds) ^ (std::get<1>(seeds) * 0xD2B74407B1CE6E93ull);
  const unsigned thresh = (unsigned)(keep_p * 16777216.f);  // keep if r < thresh
  const bool do_drop = keep_p < 1.f;

  float sv[KEEP ? ITERS : 1][8];
  for (int64_t row = (int64_t)blockIdx.x * 4 + wave; row < n_rows; row += (int64_t)gridDim.x * 4) {
    const ushort* xr = x + row * d;
    const ushort* zr = z + row * d;
    ushort* sr = s_out + row * d;
    unsigned char* mr = mask_out + row * d;

    float sum = 0.f, sq = 0.f;
    int it = 0;
    for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
      ushort8 x8 = *reinterpret_cast<const ushort8*>(xr + base);
      ushort8 z8 = *reinterpret_cast<const ushort8*>(zr + base);
      ushort8 s8;
      unsigned char m8[8];
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        unsigned char keep = 1;
        if (do_drop) keep = mix_rand24(seed, (uint64_t)row * d + base + k) < thresh;
        m8[k] = keep;
        float v = bf2f(x8[k]) + (keep ? bf2f(z8[k]) * inv_keep : 0.f);
        if (KEEP) sv[KEEP ? it : 0][k] = v;
        s8[k] = f2bf(v);
        sum += v;
        sq += v * v;
      }
      *reinterpret_cast<ushort8*>(sr + base) = s8;
#pragma unroll
      for (int k = 0; k < 8; ++k) mr[base + k] = m8[k];
    }
    sum = wave_sum(sum);
    sq = wave_sum(sq);
    const float mu = sum / d;
    const float var = sq / d - mu * mu;
    const float rs = rsqrtf(var + eps);
    if (lane == 0) {
      mean_out[row] = mu;
      rstd_out[row] = rs;
    }
    ushort* yr = y + row * d;
    it = 0;
    for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
      float vk[8];
      if (!KEEP) {
        ushort8 s8 = *reinterpret_cast<const ushort8*>(sr + base);
#pragma unroll
        for (int k = 0; k < 8; ++k) vk[k] = bf2f(s8[k]);
      } else {
#pragma unroll
        for (int k = 0; k < 8; ++k) vk[k] = sv[KEEP ? it : 0][k];
      }
      ushort8 w8 = *reinterpret_cast<const ushort8*>(w + base);
      ushort8 b8 = *reinterpret_cast<const ushort8*>(b + base);
      ushort8 o8;
#pragma unroll
      for (int k = 0; k < 8; ++k)
        o8[k] = f2bf((vk[k] - mu) * rs * bf2f(w8[k]) + bf2f(b8[k]));
      *reinterpret_cast<ushort8*>(yr + base) = o8;
    }
  }
}

// backward: dx = LN-backward(dy) over s; dz = dx * mask * inv_keep;
// dw/db accumulate per-wave into one [P][2d] partial (host folds by sum).
template <int ITERS, bool KEEP>
__global__ __launch_bounds__(256) void dropout_add_ln_bwd_t(
    const ushort* __restrict__ dy, const ushort* __restrict__ s,
    const ushort* __restrict__ w, const unsigned char* __restrict__ mask,
    const float* __restrict__ mean, const float* __restrict__ rstd,
    ushort* __restrict__ dx, ushort* __restrict__ dz,
    float* __restrict__ dwdb_partial, int64_t n_rows, int d, float inv_keep) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  float accw[ITERS][8];
  float accb[ITERS][8];
#pragma unroll
  for (int i = 0; i < ITERS; ++i)
#pragma unroll
    for (int k = 0; k < 8; ++k) { accw[i][k] = 0.f; accb[i][k] = 0.f; }

  float sv[KEEP ? ITERS : 1][8];
  float gv[KEEP ? ITERS : 1][8];

  for (int64_t row = (int64_t)blockIdx.x * 4 + wave; row < n_rows; row += (int64_t)gridDim.x * 4) {
    const ushort* dyr = dy + row * d;
    const ushort* srr = s + row * d;
    const float mu = mean[row], rs = rstd[row];

    float s1 = 0.f, s2 = 0.f;
    int it = 0;
    for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
      ushort8 g8 = *reinterpret_cast<const ushort8*>(dyr + base);
      ushort8 s8 = *reinterpret_cast<const ushort8*>(srr + base);
      ushort8 w8 = *reinterpret_cast<const ushort8*>(w + base);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        float g = bf2f(g8[k]);
        float v = bf2f(s8[k]);
        if (KEEP) { gv[KEEP ? it : 0][k] = g; sv[KEEP ? it : 0][k] = v; }
        float gw = g * bf2f(w8[k]);
        s1 += gw;
        s2 += gw * (v - mu) * rs;
      }
    }
    s1 = wave_sum(s1) / d;
    s2 = wave_sum(s2) / d;

    ushort* dxr = dx + row * d;
    ushort* dzr = dz + row * d;
    const unsigned char* mr = mask + row * d;
    it = 0;
    for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
      float gk[8], vk[8];
      if (!KEEP) {
        ushort8 g8 = *reinterpret_cast<const ushort8*>(dyr + base);
        ushort8 s8 = *reinterpret_cast<const ushort8*>(srr + base);
#pragma unroll
        for (int k = 0; k < 8; ++k) { gk[k] = bf2f(g8[k]); vk[k] = bf2f(s8[k]); }
      } else {
#pragma unroll
        for (int k = 0; k < 8; ++k) { gk[k] = gv[KEEP ? it : 0][k]; vk[k] = sv[KEEP ? it : 0][k]; }
      }
      ushort8 w8 = *reinterpret_cast<const ushort8*>(w + base);
      ushort8 ox, oz;
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        const float xhat = (vk[k] - mu) * rs;
        const float dsv = rs * (gk[k] * bf2f(w8[k]) - s1 - xhat * s2);
        ox[k] = f2bf(dsv);
        oz[k] = f2bf(mr[base + k] ? dsv * inv_keep : 0.f);
        accw[it][k] += gk[k] * xhat;
        accb[it][k] += gk[k];
      }
      *reinterpret_cast<ushort8*>(dxr + base) = ox;
      *reinterpret_cast<ushort8*>(dzr + base) = oz;
    }
  }

  float* dwp = dwdb_partial + ((int64_t)blockIdx.x * 4 + wave) * (2 * (int64_t)d);
  int it = 0;
  for (int base = lane * 8; base < d; base += 64 * 8, ++it) {
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      dwp[base + k] = accw[it][k];
      dwp[d + base + k] = accb[it][k];
    }
  }
}

extern "C" hipError_t launch_dropout_add_ln_fwd(
    const void* x, const void* z, const void* w, const void* b, void* y, void* s_out,
    void* mask_out, void* mean_out, void* rstd_out, long long n_rows, int d, float eps,
    float keep_p, float inv_keep, at::PhiloxCudaState rng, int n_blocks, hipStream_t stream) {
  dim3 g(n_blocks), blk(256);
#define DALN_F(N, KEEP) \
  hipLaunchKernelGGL((dropout_add_ln_fwd_t<N, KEEP>), g, blk, 0, stream, (const ushort*)x, \
                     (const ushort*)z, (const ushort*)w, (const ushort*)b, (ushort*)y, \
                     (ushort*)s_out, (unsigned char*)mask_out, (float*)mean_out, \
                     (float*)rstd_out, (int64_t)n_rows, d, eps, keep_p, inv_keep, rng)
  if (d <= 512) DALN_F(1, true);
  else if (d <= 1024) DALN_F(2, true);
  else if (d <= 2048) DALN_F(4, false);
  else return hipErrorInvalidValue;
#undef DALN_F
  return hipGetLastError();
}

extern "C" hipError_t launch_dropout_add_ln_bwd(
    const void* dy, const void* s, const void* w, const void* mask, const void* mean,
    const void* rstd, void* dx, void* dz, void* dwdb_partial, long long n_rows, int d,
    float inv_keep, int n_blocks, hipStream_t stream) {
  dim3 g(n_blocks), blk(256);
#define DALN_B(N, KEEP) \
  hipLaunchKernelGGL((dropout_add_ln_bwd_t<N, KEEP>), g, blk, 0, stream, (const ushort*)dy, \
                     (const ushort*)s, (const ushort*)w, (const unsigned char*)mask, \
                     (const float*)mean, (const float*)rstd, (ushort*)dx, (ushort*)dz, \
                     (float*)dwdb_partial, (int64_t)n_rows, d, inv_keep)
  if (d <= 512) DALN_B(1, true);
  else if (d <= 1024) DALN_B(2, true);
  else if (d <= 2048) DALN_B(4, false);
  else return hipErrorInvalidValue;
#undef DALN_B
  return hipGetLastError();
}
