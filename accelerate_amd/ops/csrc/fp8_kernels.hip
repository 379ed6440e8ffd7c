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

template <typename FP8>
__device__ __forceinline__ void fp8_cast_amax_body(const ushort* __restrict__ in_bf16,
                                                   unsigned char* __restrict__ out,
                                                   const float* __restrict__ scale,
                                                   float* __restrict__ amax_out,
                                                   int64_t n) {
  const float s = *scale;
  float local_amax = 0.f;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x * 8;
  // 8 bf16 per thread (16 B/lane loads — CDNA4 Guideline 13)
  for (int64_t base = ((int64_t)blockIdx.x * blockDim.x + threadIdx.x) * 8; base < n; base += stride) {
    if (base + 7 < n) {
      ushort4 a = *reinterpret_cast<const ushort4*>(in_bf16 + base);
      ushort4 b = *reinterpret_cast<const ushort4*>(in_bf16 + base + 4);
      unsigned char r[8];
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        ushort u = (k < 4) ? (&a.x)[k] : (&b.x)[k - 4];
        float x = __bfloat162float(*reinterpret_cast<__hip_bfloat16*>(&u));
        local_amax = fmaxf(local_amax, fabsf(x));
        FP8 q(x * s);
        r[k] = *reinterpret_cast<unsigned char*>(&q);
      }
      *reinterpret_cast<uint2*>(out + base) = *reinterpret_cast<uint2*>(r);
    } else {
      for (int64_t j = base; j < n; ++j) {
        ushort u = in_bf16[j];
        float x = __bfloat162float(*reinterpret_cast<__hip_bfloat16*>(&u));
        local_amax = fmaxf(local_amax, fabsf(x));
        FP8 q(x * s);
        out[j] = *reinterpret_cast<unsigned char*>(&q);
      }
    }
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
  fp8_cast_amax_body<__hip_fp8_e4m3>(in, out, scale, amax, n);
}

__global__ void fp8_cast_amax_e5m2(const ushort* in, unsigned char* out,
                                   const float* scale, float* amax, int64_t n) {
  fp8_cast_amax_body<__hip_fp8_e5m2>(in, out, scale, amax, n);
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
