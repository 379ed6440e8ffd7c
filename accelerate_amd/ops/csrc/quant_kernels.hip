// CDNA4 weight-only quantization kernels (gfx950).
//
// MI355X-native replacement for the reference's bitsandbytes integration
// (reference utils/bnb.py:44-199 load_and_quantize_model / :280
// replace_with_bnb_layers — CUDA-only upstream). Storage formats:
//   int8: per-output-channel symmetric scale, q in [-127,127]
//   int4: group-wise symmetric scale (group g along in-features),
//         offset-binary nibbles (stored = q+8, q in [-8,7]), 2 per byte,
//         element 2k in the LOW nibble of byte k.
//
// Dequant is the hot op (every forward touches it); both kernels move
// 16 output elements per lane per iteration (Guideline 13: 16 B loads /
// 32 B bf16 stores) so they run at HBM streaming rate. The w8a16 GEMV
// fuses dequant into a skinny matvec for decode shapes where writing a
// dequantized weight matrix back to HBM would double the traffic.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

typedef ushort ushort8 __attribute__((ext_vector_type(8)));  // 16 B = 8 bf16
typedef char char16_t_v __attribute__((ext_vector_type(16)));

__device__ __forceinline__ ushort f32_to_bf16_rne(float f) {
  unsigned int u = __float_as_uint(f);
  u += 0x7fff + ((u >> 16) & 1);  // round-to-nearest-even
  return (ushort)(u >> 16);
}

// int8 [rows, cols] * scale[rows] -> bf16 [rows, cols]; cols % 16 == 0 so a
// 16-element chunk never crosses a row (one scale per chunk).
__global__ void int8_dequant_kernel(const char* __restrict__ q,
                                    const float* __restrict__ scale,
                                    ushort* __restrict__ out,
                                    int64_t rows, int64_t cols) {
  const int64_t nvec = (rows * cols) >> 4;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    const int64_t base = v << 4;
    const float s = scale[base / cols];
    char16_t_v b = *reinterpret_cast<const char16_t_v*>(q + base);
    ushort8 lo, hi;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      lo[k] = f32_to_bf16_rne((float)b[k] * s);
      hi[k] = f32_to_bf16_rne((float)b[k + 8] * s);
    }
    *reinterpret_cast<ushort8*>(out + base) = lo;
    *reinterpret_cast<ushort8*>(out + base + 8) = hi;
  }
}

// packed int4 [rows, cols/2] * scale[rows, cols/group] -> bf16 [rows, cols];
// cols % 16 == 0 and group % 16 == 0 so one 16-element chunk has one scale.
__global__ void int4_dequant_kernel(const unsigned char* __restrict__ q,
                                    const float* __restrict__ scale,
                                    ushort* __restrict__ out,
                                    int64_t rows, int64_t cols, int group) {
  const int64_t nvec = (rows * cols) >> 4;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  const int64_t groups_per_row = cols / group;
  for (int64_t v = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; v < nvec; v += stride) {
    const int64_t base = v << 4;                    // element index
    const int64_t row = base / cols;
    const int64_t col = base - row * cols;
    const float s = scale[row * groups_per_row + col / group];
    uint2 packed = *reinterpret_cast<const uint2*>(q + (base >> 1));  // 8 B = 16 nibbles
    const unsigned char* pb = reinterpret_cast<const unsigned char*>(&packed);
    ushort8 lo, hi;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      lo[2 * k] = f32_to_bf16_rne((float)((int)(pb[k] & 0xF) - 8) * s);
      lo[2 * k + 1] = f32_to_bf16_rne((float)((int)(pb[k] >> 4) - 8) * s);
      hi[2 * k] = f32_to_bf16_rne((float)((int)(pb[k + 4] & 0xF) - 8) * s);
      hi[2 * k + 1] = f32_to_bf16_rne((float)((int)(pb[k + 4] >> 4) - 8) * s);
    }
    *reinterpret_cast<ushort8*>(out + base) = lo;
    *reinterpret_cast<ushort8*>(out + base + 8) = hi;
  }
}

// y[b, r] = (sum_k q[r,k] * x[b,k]) * scale[r] + bias[r]
// Decode-shaped (batch <= 8): memory-bound on the int8 weight stream, so
// fusing dequant into the matvec reads each weight byte exactly once.
// One wave per output row, 4 rows per block; lane loads 16 int8 weights
// (16 B) + 16 bf16 activations (32 B) per iteration. cols % 1024 == 0 for
// the fast path (hidden sizes are); host falls back to dequant+GEMM else.
__global__ void w8a16_gemv_kernel(const char* __restrict__ q,
                                  const float* __restrict__ scale,
                                  const ushort* __restrict__ x,
                                  const ushort* __restrict__ bias,
                                  ushort* __restrict__ y,
                                  int64_t rows, int64_t cols, int batch) {
  const int wave = threadIdx.x >> 6;               // 0..3
  const int lane = threadIdx.x & 63;
  const int64_t row = (int64_t)blockIdx.x * 4 + wave;
  if (row >= rows) return;
  const char* wrow = q + row * cols;
  for (int b = 0; b < batch; ++b) {
    const ushort* xb = x + (int64_t)b * cols;
    // two independent accumulators + 2x unroll: the single-acc version's
    // 16-deep FMA chain per 16 B made the loop latency-bound (~1.9 TB/s)
    float acc0 = 0.f, acc1 = 0.f;
#pragma unroll 2
    for (int64_t k0 = (int64_t)lane * 16; k0 < cols; k0 += 64 * 16) {
      char16_t_v w16 = *reinterpret_cast<const char16_t_v*>(wrow + k0);
      ushort8 xa = *reinterpret_cast<const ushort8*>(xb + k0);
      ushort8 xc = *reinterpret_cast<const ushort8*>(xb + k0 + 8);
#pragma unroll
      for (int k = 0; k < 8; ++k) {
        ushort ua = xa[k], uc = xc[k];
        acc0 += (float)w16[k] * __bfloat162float(*reinterpret_cast<__hip_bfloat16*>(&ua));
        acc1 += (float)w16[k + 8] * __bfloat162float(*reinterpret_cast<__hip_bfloat16*>(&uc));
      }
    }
    float acc = acc0 + acc1;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
    if (lane == 0) {
      float r = acc * scale[row];
      if (bias != nullptr) {
        ushort ub = bias[row];
        r += __bfloat162float(*reinterpret_cast<__hip_bfloat16*>(&ub));
      }
      y[(int64_t)b * rows + row] = f32_to_bf16_rne(r);
    }
  }
}
