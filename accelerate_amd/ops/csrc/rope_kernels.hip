// Fused rotary position embedding (half-split layout) for CDNA4.
//
// One kernel replaces torch's slice + 4 muls + 2 adds + cat chain (~7
// kernels, 6× the memory traffic). bf16 activations, fp32 cos/sin tables
// (host-precomputed — CDNA4 guide: on-device trig turns memory-bound ops
// VALU-bound). Backward is rotation by −θ (pass negate_sin).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

typedef ushort ushort8r __attribute__((ext_vector_type(8)));

__device__ __forceinline__ float rbf2f(ushort u) {
  union { unsigned int i; float f; } cv;
  cv.i = ((unsigned int)u) << 16;
  return cv.f;
}

__device__ __forceinline__ ushort rf2bf(float f) {
  union { float f; unsigned int i; } cv;
  cv.f = f;
  if ((cv.i & 0x7F800000u) == 0x7F800000u) return (ushort)(cv.i >> 16) | (ushort)((cv.i & 0xFFFFu) ? 0x40 : 0);
  cv.i += 0x7FFFu + ((cv.i >> 16) & 1u);
  return (ushort)(cv.i >> 16);
}

}  // namespace

// x,y: [BH, S, D] contiguous bf16; cos/sin: [>=S, D/2] fp32 (pre-offset).
// Each thread rotates 8 pairs: loads 16 B from each half.
__global__ void rope_bf16_kernel(const ushort* __restrict__ x, ushort* __restrict__ y,
                                 const float* __restrict__ cos_t, const float* __restrict__ sin_t,
                                 int64_t n_bh, int S, int D, float sin_sign) {
  const int half = D >> 1;
  const int chunks_per_row = half >> 3;  // 8-pair chunks per (bh, s)
  const int64_t total = n_bh * S * chunks_per_row;
  for (int64_t idx = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; idx < total;
       idx += (int64_t)gridDim.x * blockDim.x) {
    const int c8 = (int)(idx % chunks_per_row);
    const int64_t row = idx / chunks_per_row;  // bh*S + s
    const int s = (int)(row % S);
    const int d0 = c8 * 8;
    const ushort* x1p = x + row * D + d0;
    const ushort* x2p = x1p + half;
    ushort8r x1 = *reinterpret_cast<const ushort8r*>(x1p);
    ushort8r x2 = *reinterpret_cast<const ushort8r*>(x2p);
    float4 c0 = *reinterpret_cast<const float4*>(cos_t + (int64_t)s * half + d0);
    float4 c1 = *reinterpret_cast<const float4*>(cos_t + (int64_t)s * half + d0 + 4);
    float4 s0 = *reinterpret_cast<const float4*>(sin_t + (int64_t)s * half + d0);
    float4 s1 = *reinterpret_cast<const float4*>(sin_t + (int64_t)s * half + d0 + 4);
    ushort8r y1, y2;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float a = rbf2f(x1[k]);
      float b = rbf2f(x2[k]);
      float ck = (k < 4) ? (&c0.x)[k] : (&c1.x)[k - 4];
      float sk = ((k < 4) ? (&s0.x)[k] : (&s1.x)[k - 4]) * sin_sign;
      y1[k] = rf2bf(a * ck - b * sk);
      y2[k] = rf2bf(b * ck + a * sk);
    }
    *reinterpret_cast<ushort8r*>(y + row * D + d0) = y1;
    *reinterpret_cast<ushort8r*>(y + row * D + d0 + half) = y2;
  }
}
