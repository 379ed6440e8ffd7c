// CDNA4 fp8 (OCP e4m3fn / e5m2) cast kernels with fused amax capture.
//
// Delayed-scaling recipe: the cast uses the PREVIOUS step's scale while the
// SAME pass records the current amax (one HBM sweep instead of the two a
// separate abs().max() would need). gfx950 fp8 is OCP (e4m3fn/e5m2), NOT the
// MI300X fnuz variant (CDNA4 guide §4).

#include "multi_tensor.h"

#include <hip/hip_runtime.h>
#include <hip/hip_fp16.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp8.h>

template <bool kE5M2, bool kWord>
__device__ __forceinline__ unsigned int cvt_pk2(float lo, float hi, unsigned int old) {
  // hardware packed fp32x2 -> fp8x2 convert (v_cvt_pk_fp8/bf8_f32). The
  // convert does NOT saturate and OCP e4m3fn has no inf encoding — overflow
  // becomes NaN — so clamp to the format max first (delayed scaling can
  // transiently under-scale between amax refreshes).
  constexpr float kMax = kE5M2 ? 57344.f : 448.f;
  lo = fminf(fmaxf(lo, -kMax), kMax);
  hi = fminf(fmaxf(hi, -kMax), kMax);
  if constexpr (kE5M2) return __builtin_amdgcn_cvt_pk_bf8_f32(lo, hi, old, kWord);
  else return __builtin_amdgcn_cvt_pk_fp8_f32(lo, hi, old, kWord);
}

typedef ushort ushort8 __attribute__((ext_vector_type(8)));  // 16 B = 8 bf16

template <bool kE5M2>
__device__ __forceinline__ unsigned int cvt8(const float* x, float s, float& amax, int o) {
  amax = fmaxf(amax, fmaxf(fmaxf(fabsf(x[o]), fabsf(x[o + 1])), fmaxf(fabsf(x[o + 2]), fabsf(x[o + 3]))));
  unsigned int q = cvt_pk2<kE5M2, false>(x[o] * s, x[o + 1] * s, 0u);
  return cvt_pk2<kE5M2, true>(x[o + 2] * s, x[o + 3] * s, q);
}

template <typename FP8, bool kE5M2>
__device__ __forceinline__ void fp8_cast_amax_body(const ushort* __restrict__ in_bf16,
                                                   unsigned char* __restrict__ out,
                                                   const float* __restrict__ scale,
                                                   float* __restrict__ amax_out,
                                                   int64_t n) {
  const float s = *scale;
  float local_amax = 0.f;
  // 16 bf16 per thread: 2×16 B loads, 1×16 B store (true 16 B/lane —
  // ushort4 is only 8 B; CDNA4 Guideline 13) + gfx950 packed converts
  const int64_t nvec = n >> 4;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    const int64_t base = v << 4;
    ushort8 a = *reinterpret_cast<const ushort8*>(in_bf16 + base);
    ushort8 b = *reinterpret_cast<const ushort8*>(in_bf16 + base + 8);
    float x[16];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      ushort ua = a[k], ub = b[k];
      x[k] = __bfloat162float(*reinterpret_cast<__hip_bfloat16*>(&ua));
      x[k + 8] = __bfloat162float(*reinterpret_cast<__hip_bfloat16*>(&ub));
    }
    uint4 r;
    r.x = cvt8<kE5M2>(x, s, local_amax, 0);
    r.y = cvt8<kE5M2>(x, s, local_amax, 4);
    r.z = cvt8<kE5M2>(x, s, local_amax, 8);
    r.w = cvt8<kE5M2>(x, s, local_amax, 12);
    *reinterpret_cast<uint4*>(out + base) = r;
  }
  // tail
  for (int64_t j = (nvec << 4) + blockIdx.x * blockDim.x + threadIdx.x; j < n;
       j += (int64_t)gridDim.x * blockDim.x) {
    ushort u = in_bf16[j];
    float x = __bfloat162float(*reinterpret_cast<__hip_bfloat16*>(&u));
    local_amax = fmaxf(local_amax, fabsf(x));
    FP8 q(x * s);
    out[j] = *reinterpret_cast<unsigned char*>(&q);
  }
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) local_amax = fmaxf(local_amax, __shfl_down(local_amax, off, 64));
  __shared__ float wave_max[kBlockThreads / 64];
  const int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) wave_max[wave] = local_amax;
  __syncthreads();
  if (threadIdx.x == 0) {
    float m = 0.f;
#pragma unroll
    for (int w = 0; w < kBlockThreads / 64; ++w) m = fmaxf(m, wave_max[w]);
    // |m| >= 0 so the float bit pattern is monotonic as unsigned int
    atomicMax(reinterpret_cast<unsigned int*>(amax_out), __float_as_uint(m));
  }
}

__global__ void fp8_cast_amax_e4m3(const ushort* in, unsigned char* out,
                                   const float* scale, float* amax, int64_t n) {
  fp8_cast_amax_body<__hip_fp8_e4m3, false>(in, out, scale, amax, n);
}

__global__ void fp8_cast_amax_e5m2(const ushort* in, unsigned char* out,
                                   const float* scale, float* amax, int64_t n) {
  fp8_cast_amax_body<__hip_fp8_e5m2, true>(in, out, scale, amax, n);
}

// ---------------------------------------------------------------------------
// fused cast + transpose + amax: bf16 [R,C] -> fp8 [R,C] AND fp8 [C,R].
// Every fp8 GEMM operand needs both layouts (fwd/dgrad use X, W; wgrad uses
// X^T, G^T) — one HBM read produces both (TE's cast_transpose pattern).
// 64×64 tiles staged through LDS (padded rows, Guideline 4). R,C % 64 == 0.
// ---------------------------------------------------------------------------

template <bool kE5M2>
__device__ __forceinline__ void cast_transpose_body(const ushort* __restrict__ in,
                                                    unsigned char* __restrict__ out,
                                                    unsigned char* __restrict__ out_t,
                                                    const float* __restrict__ scale,
                                                    float* __restrict__ amax_out,
                                                    int64_t R, int64_t C) {
  // 128×128 tile: every out AND out_t row receives a FULL 128 B cache line
  // per block (64-wide tiles left half-line writes → L2 read-modify-write).
  constexpr int kTile = 128;
  constexpr int kPad = 8;  // LDS row stride 136 B breaks power-of-2 conflicts
  __shared__ unsigned char lds[kTile][kTile + kPad];
  const float s = *scale;
  const int64_t tile_c = (int64_t)blockIdx.x * kTile;
  const int64_t tile_r = (int64_t)blockIdx.y * kTile;
  const int t = threadIdx.x;  // 256 threads
  float local_amax = 0.f;

  // phase 1: 4 iterations × one 16-bf16 chunk (2×16 B loads, 16 B stores)
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    const int task = t + it * 256;        // 1024 chunks
    const int r = task >> 3;              // 0..127
    const int cchunk = (task & 7) * 16;   // 0..112
    const ushort* src = in + (tile_r + r) * C + tile_c + cchunk;
    ushort8 a = *reinterpret_cast<const ushort8*>(src);
    ushort8 b = *reinterpret_cast<const ushort8*>(src + 8);
    float x[16];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      ushort ua = a[k], ub = b[k];
      x[k] = __bfloat162float(*reinterpret_cast<__hip_bfloat16*>(&ua));
      x[k + 8] = __bfloat162float(*reinterpret_cast<__hip_bfloat16*>(&ub));
    }
    uint4 q;
    q.x = cvt8<kE5M2>(x, s, local_amax, 0);
    q.y = cvt8<kE5M2>(x, s, local_amax, 4);
    q.z = cvt8<kE5M2>(x, s, local_amax, 8);
    q.w = cvt8<kE5M2>(x, s, local_amax, 12);
    *reinterpret_cast<uint4*>(out + (tile_r + r) * C + tile_c + cchunk) = q;
    *reinterpret_cast<uint4*>(&lds[r][cchunk]) = q;
  }
  __syncthreads();

  // phase 2: transpose — thread gathers 16 rows of one column per iteration
#pragma unroll
  for (int it = 0; it < 4; ++it) {
    const int task = t + it * 256;        // 1024 chunks
    const int c = task >> 3;              // source column = out_t row
    const int rchunk = (task & 7) * 16;
    unsigned char bytes[16];
#pragma unroll
    for (int k = 0; k < 16; ++k) bytes[k] = lds[rchunk + k][c];
    *reinterpret_cast<uint4*>(out_t + (tile_c + c) * R + tile_r + rchunk) =
        *reinterpret_cast<uint4*>(bytes);
  }

#pragma unroll
  for (int off = 32; off > 0; off >>= 1) local_amax = fmaxf(local_amax, __shfl_down(local_amax, off, 64));
  __shared__ float wave_max[4];
  if ((t & 63) == 0) wave_max[t >> 6] = local_amax;
  __syncthreads();
  if (t == 0) {
    float m = fmaxf(fmaxf(wave_max[0], wave_max[1]), fmaxf(wave_max[2], wave_max[3]));
    atomicMax(reinterpret_cast<unsigned int*>(amax_out), __float_as_uint(m));
  }
}

__global__ void fp8_cast_transpose_e4m3(const ushort* in, unsigned char* out, unsigned char* out_t,
                                        const float* scale, float* amax, int64_t R, int64_t C) {
  cast_transpose_body<false>(in, out, out_t, scale, amax, R, C);
}

__global__ void fp8_cast_transpose_e5m2(const ushort* in, unsigned char* out, unsigned char* out_t,
                                        const float* scale, float* amax, int64_t R, int64_t C) {
  cast_transpose_body<true>(in, out, out_t, scale, amax, R, C);
}

// scale update: scale = fp8_max / (max(history) * 2^margin), guarded for 0
__global__ void fp8_update_scale(const float* __restrict__ history, int hist_len,
                                 float fp8_max, float margin_pow2,
                                 float* __restrict__ scale,
                                 float* __restrict__ scale_inv) {
  float m = 0.f;
  for (int i = 0; i < hist_len; ++i) m = fmaxf(m, history[i]);
  float s = (m > 0.f) ? fp8_max / (m * margin_pow2) : 1.f;
  *scale = s;
  *scale_inv = 1.f / s;
}
