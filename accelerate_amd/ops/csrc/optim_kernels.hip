// CDNA4 (gfx950) optimizer-path kernels: fused AdamW, global L2 norm,
// clip-coef + multi-tensor scale, unscale + non-finite check.
//
// All kernels are HBM-bound sweeps: float4 main loop (16 B/lane coalesced),
// scalar tail; grid = total chunks across all tensors (≫256 workgroups on
// real models, filling all 8 XCDs).

#include "multi_tensor.h"

#include <hip/hip_runtime.h>
#include <cmath>

// ---------------------------------------------------------------------------
// fused AdamW (decoupled weight decay), fp32 params/grads/state.
// Math mirrors torch.optim.AdamW exactly so numerics tests can compare:
//   p *= (1 - lr*wd)
//   m = b1*m + (1-b1)*g ; v = b2*v + (1-b2)*g²
//   p -= (lr/bc1) * m / (sqrt(v)/sqrt(bc2) + eps)
// ---------------------------------------------------------------------------

__global__ void fused_adamw_kernel(
    TensorListMeta meta,
    float lr, float beta1, float beta2, float eps, float weight_decay,
    float bias_correction1, float rsqrt_bias_correction2,
    const float* __restrict__ grad_scale,   // nullptr or *inv_scale applied to grads
    const float* __restrict__ found_inf) {  // nullptr or skip-step flag
  if (found_inf != nullptr && *found_inf != 0.f) return;

  const int cid = blockIdx.x;
  const int t = find_tensor(meta.chunk_prefix, meta.n_tensors, cid);
  const int64_t chunk_in_tensor = cid - meta.chunk_prefix[t];
  const int64_t numel = meta.numels[t];
  const int64_t lo = chunk_in_tensor * kChunkSize;
  const int64_t hi = min(lo + kChunkSize, numel);

  float* __restrict__ p = reinterpret_cast<float*>(meta.addrs[0 * meta.n_tensors + t]);
  const float* __restrict__ g = reinterpret_cast<const float*>(meta.addrs[1 * meta.n_tensors + t]);
  float* __restrict__ m = reinterpret_cast<float*>(meta.addrs[2 * meta.n_tensors + t]);
  float* __restrict__ v = reinterpret_cast<float*>(meta.addrs[3 * meta.n_tensors + t]);

  const float step_size = lr / bias_correction1;
  const float decay = 1.f - lr * weight_decay;
  const float gscale = (grad_scale != nullptr) ? *grad_scale : 1.f;

  const int64_t tid = threadIdx.x;
  // float4 main loop over [lo, hi)
  int64_t i = lo + tid * 4;
  const int64_t vec_end = lo + ((hi - lo) & ~int64_t(3));
  for (; i + 3 < vec_end; i += kBlockThreads * 4) {
    float4 gp = *reinterpret_cast<const float4*>(g + i);
    float4 pp = *reinterpret_cast<float4*>(p + i);
    float4 mp = *reinterpret_cast<float4*>(m + i);
    float4 vp = *reinterpret_cast<float4*>(v + i);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float gk = (&gp.x)[k] * gscale;
      float pk = (&pp.x)[k] * decay;
      float mk = beta1 * (&mp.x)[k] + (1.f - beta1) * gk;
      float vk = beta2 * (&vp.x)[k] + (1.f - beta2) * gk * gk;
      float denom = sqrtf(vk) * rsqrt_bias_correction2 + eps;
      pk -= step_size * mk / denom;
      (&pp.x)[k] = pk; (&mp.x)[k] = mk; (&vp.x)[k] = vk;
    }
    *reinterpret_cast<float4*>(p + i) = pp;
    *reinterpret_cast<float4*>(m + i) = mp;
    *reinterpret_cast<float4*>(v + i) = vp;
  }
  // scalar tail
  for (int64_t j = vec_end + tid; j < hi; j += kBlockThreads) {
    float gk = g[j] * gscale;
    float pk = p[j] * decay;
    float mk = beta1 * m[j] + (1.f - beta1) * gk;
    float vk = beta2 * v[j] + (1.f - beta2) * gk * gk;
    float denom = sqrtf(vk) * rsqrt_bias_correction2 + eps;
    pk -= step_size * mk / denom;
    p[j] = pk; m[j] = mk; v[j] = vk;
  }
}

// ---------------------------------------------------------------------------
// hipGraph-capturable AdamW: step and lr live in DEVICE memory so a captured
// graph replays with fresh values (host writes lr before replay; the prep
// kernel increments step in-graph). Bias corrections are computed per block
// from *step — two powf per workgroup, noise.
// ---------------------------------------------------------------------------

__global__ void adamw_incr_step_kernel(float* __restrict__ step) { *step += 1.f; }

__global__ void fused_adamw_dev_kernel(
    TensorListMeta meta,
    const float* __restrict__ step_ptr, const float* __restrict__ lr_ptr,
    float beta1, float beta2, float eps, float weight_decay,
    const float* __restrict__ grad_scale,
    const float* __restrict__ found_inf) {
  if (found_inf != nullptr && *found_inf != 0.f) return;

  const float step_f = *step_ptr;
  const float lr = *lr_ptr;
  const float bias_correction1 = 1.f - powf(beta1, step_f);
  const float rsqrt_bias_correction2 = rsqrtf(1.f - powf(beta2, step_f));

  const int cid = blockIdx.x;
  const int t = find_tensor(meta.chunk_prefix, meta.n_tensors, cid);
  const int64_t chunk_in_tensor = cid - meta.chunk_prefix[t];
  const int64_t numel = meta.numels[t];
  const int64_t lo = chunk_in_tensor * kChunkSize;
  const int64_t hi = min(lo + kChunkSize, numel);

  float* __restrict__ p = reinterpret_cast<float*>(meta.addrs[0 * meta.n_tensors + t]);
  const float* __restrict__ g = reinterpret_cast<const float*>(meta.addrs[1 * meta.n_tensors + t]);
  float* __restrict__ m = reinterpret_cast<float*>(meta.addrs[2 * meta.n_tensors + t]);
  float* __restrict__ v = reinterpret_cast<float*>(meta.addrs[3 * meta.n_tensors + t]);

  const float step_size = lr / bias_correction1;
  const float decay = 1.f - lr * weight_decay;
  const float gscale = (grad_scale != nullptr) ? *grad_scale : 1.f;

  const int64_t tid = threadIdx.x;
  int64_t i = lo + tid * 4;
  const int64_t vec_end = lo + ((hi - lo) & ~int64_t(3));
  for (; i + 3 < vec_end; i += kBlockThreads * 4) {
    float4 gp = *reinterpret_cast<const float4*>(g + i);
    float4 pp = *reinterpret_cast<float4*>(p + i);
    float4 mp = *reinterpret_cast<float4*>(m + i);
    float4 vp = *reinterpret_cast<float4*>(v + i);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float gk = (&gp.x)[k] * gscale;
      float pk = (&pp.x)[k] * decay;
      float mk = beta1 * (&mp.x)[k] + (1.f - beta1) * gk;
      float vk = beta2 * (&vp.x)[k] + (1.f - beta2) * gk * gk;
      float denom = sqrtf(vk) * rsqrt_bias_correction2 + eps;
      pk -= step_size * mk / denom;
      (&pp.x)[k] = pk; (&mp.x)[k] = mk; (&vp.x)[k] = vk;
    }
    *reinterpret_cast<float4*>(p + i) = pp;
    *reinterpret_cast<float4*>(m + i) = mp;
    *reinterpret_cast<float4*>(v + i) = vp;
  }
  for (int64_t j = vec_end + tid; j < hi; j += kBlockThreads) {
    float gk = g[j] * gscale;
    float pk = p[j] * decay;
    float mk = beta1 * m[j] + (1.f - beta1) * gk;
    float vk = beta2 * v[j] + (1.f - beta2) * gk * gk;
    float denom = sqrtf(vk) * rsqrt_bias_correction2 + eps;
    pk -= step_size * mk / denom;
    p[j] = pk; m[j] = mk; v[j] = vk;
  }
}

// ---------------------------------------------------------------------------
// bf16-weights AdamW: model params/grads are bf16, the optimizer keeps the
// fp32 master copy + fp32 m/v (standard production bf16 recipe; kills the
// per-step autocast weight-cast traffic and halves gradient all-reduce bytes
// over xGMI). Lists: [0]=param_bf16, [1]=grad_bf16, [2]=m, [3]=v, [4]=master.
// ---------------------------------------------------------------------------

#include <hip/hip_bf16.h>

__device__ __forceinline__ float bf16_to_f32(ushort u) {
  union { unsigned int i; float f; } cv;
  cv.i = ((unsigned int)u) << 16;
  return cv.f;
}

__device__ __forceinline__ ushort f32_to_bf16(float f) {
  // round-to-nearest-even, matching torch's fp32->bf16 cast exactly
  union { float f; unsigned int i; } cv;
  cv.f = f;
  if ((cv.i & 0x7F800000u) == 0x7F800000u) {  // inf/nan: truncate
    return (ushort)(cv.i >> 16) | (ushort)((cv.i & 0xFFFFu) ? 0x40 : 0);
  }
  unsigned int lsb = (cv.i >> 16) & 1u;
  cv.i += 0x7FFFu + lsb;
  return (ushort)(cv.i >> 16);
}

__global__ void fused_adamw_bf16_kernel(
    TensorListMeta meta,
    const float* __restrict__ step_ptr, const float* __restrict__ lr_ptr,
    float beta1, float beta2, float eps, float weight_decay,
    const float* __restrict__ grad_scale,
    const float* __restrict__ found_inf) {
  if (found_inf != nullptr && *found_inf != 0.f) return;

  const float step_f = *step_ptr;
  const float lr = *lr_ptr;
  const float bias_correction1 = 1.f - powf(beta1, step_f);
  const float rsqrt_bias_correction2 = rsqrtf(1.f - powf(beta2, step_f));

  const int cid = blockIdx.x;
  const int t = find_tensor(meta.chunk_prefix, meta.n_tensors, cid);
  const int64_t chunk_in_tensor = cid - meta.chunk_prefix[t];
  const int64_t numel = meta.numels[t];
  const int64_t lo = chunk_in_tensor * kChunkSize;
  const int64_t hi = min(lo + kChunkSize, numel);

  ushort* __restrict__ p16 = reinterpret_cast<ushort*>(meta.addrs[0 * meta.n_tensors + t]);
  const ushort* __restrict__ g16 = reinterpret_cast<const ushort*>(meta.addrs[1 * meta.n_tensors + t]);
  float* __restrict__ m = reinterpret_cast<float*>(meta.addrs[2 * meta.n_tensors + t]);
  float* __restrict__ v = reinterpret_cast<float*>(meta.addrs[3 * meta.n_tensors + t]);
  float* __restrict__ w = reinterpret_cast<float*>(meta.addrs[4 * meta.n_tensors + t]);

  const float step_size = lr / bias_correction1;
  const float decay = 1.f - lr * weight_decay;
  const float gscale = (grad_scale != nullptr) ? *grad_scale : 1.f;

  const int64_t tid = threadIdx.x;
  int64_t i = lo + tid * 4;
  const int64_t vec_end = lo + ((hi - lo) & ~int64_t(3));
  for (; i + 3 < vec_end; i += kBlockThreads * 4) {
    ushort4 gp = *reinterpret_cast<const ushort4*>(g16 + i);
    float4 wp = *reinterpret_cast<float4*>(w + i);
    float4 mp = *reinterpret_cast<float4*>(m + i);
    float4 vp = *reinterpret_cast<float4*>(v + i);
    ushort4 po;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float gk = bf16_to_f32((&gp.x)[k]) * gscale;
      float pk = (&wp.x)[k] * decay;
      float mk = beta1 * (&mp.x)[k] + (1.f - beta1) * gk;
      float vk = beta2 * (&vp.x)[k] + (1.f - beta2) * gk * gk;
      float denom = sqrtf(vk) * rsqrt_bias_correction2 + eps;
      pk -= step_size * mk / denom;
      (&wp.x)[k] = pk; (&mp.x)[k] = mk; (&vp.x)[k] = vk;
      (&po.x)[k] = f32_to_bf16(pk);
    }
    *reinterpret_cast<float4*>(w + i) = wp;
    *reinterpret_cast<float4*>(m + i) = mp;
    *reinterpret_cast<float4*>(v + i) = vp;
    *reinterpret_cast<ushort4*>(p16 + i) = po;
  }
  for (int64_t j = vec_end + tid; j < hi; j += kBlockThreads) {
    float gk = bf16_to_f32(g16[j]) * gscale;
    float pk = w[j] * decay;
    float mk = beta1 * m[j] + (1.f - beta1) * gk;
    float vk = beta2 * v[j] + (1.f - beta2) * gk * gk;
    float denom = sqrtf(vk) * rsqrt_bias_correction2 + eps;
    pk -= step_size * mk / denom;
    w[j] = pk; m[j] = mk; v[j] = vk;
    p16[j] = f32_to_bf16(pk);
  }
}

// ---------------------------------------------------------------------------
// multi-tensor gather/scatter between parameter gradients and a flat bucket
// (the DDP reducer's bucket staging — one launch per bucket instead of one
// copy kernel per parameter). Dtype-agnostic byte copy, uint4-vectorized;
// list0 = tensor ptrs, list1 = flat+offset ptrs (both byte addresses).
// chunk_prefix is over BYTES here (kChunkSize-byte chunks).
// ---------------------------------------------------------------------------

__global__ void multi_tensor_copy_kernel(TensorListMeta meta, bool to_flat) {
  const int cid = blockIdx.x;
  const int t = find_tensor(meta.chunk_prefix, meta.n_tensors, cid);
  const int64_t chunk_in_tensor = cid - meta.chunk_prefix[t];
  const int64_t nbytes = meta.numels[t];
  const int64_t lo = chunk_in_tensor * kChunkSize;
  const int64_t hi = min(lo + kChunkSize, nbytes);
  const unsigned char* src = reinterpret_cast<const unsigned char*>(
      to_flat ? meta.addrs[t] : meta.addrs[meta.n_tensors + t]);
  unsigned char* dst = reinterpret_cast<unsigned char*>(
      to_flat ? meta.addrs[meta.n_tensors + t] : meta.addrs[t]);

  const int64_t tid = threadIdx.x;
  int64_t i = lo + tid * 16;
  const int64_t vec_end = lo + ((hi - lo) & ~int64_t(15));
  for (; i + 15 < vec_end; i += kBlockThreads * 16) {
    *reinterpret_cast<uint4*>(dst + i) = *reinterpret_cast<const uint4*>(src + i);
  }
  for (int64_t j = vec_end + tid; j < hi; j += kBlockThreads) dst[j] = src[j];
}

// ---------------------------------------------------------------------------
// global L2 norm²: per-wave shuffle reduce → per-block LDS reduce → one
// device-scope atomicAdd per block (Guideline 12).
// ---------------------------------------------------------------------------

__global__ void l2norm_squared_kernel(TensorListMeta meta, float* __restrict__ out) {
  const int cid = blockIdx.x;
  const int t = find_tensor(meta.chunk_prefix, meta.n_tensors, cid);
  const int64_t chunk_in_tensor = cid - meta.chunk_prefix[t];
  const int64_t numel = meta.numels[t];
  const int64_t lo = chunk_in_tensor * kChunkSize;
  const int64_t hi = min(lo + kChunkSize, numel);
  const float* __restrict__ g = reinterpret_cast<const float*>(meta.addrs[t]);

  float acc = 0.f;
  const int64_t tid = threadIdx.x;
  int64_t i = lo + tid * 4;
  const int64_t vec_end = lo + ((hi - lo) & ~int64_t(3));
  for (; i + 3 < vec_end; i += kBlockThreads * 4) {
    float4 gp = *reinterpret_cast<const float4*>(g + i);
    acc += gp.x * gp.x + gp.y * gp.y + gp.z * gp.z + gp.w * gp.w;
  }
  for (int64_t j = vec_end + tid; j < hi; j += kBlockThreads) acc += g[j] * g[j];

  // wave64 shuffle reduce
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) acc += __shfl_down(acc, off, 64);
  __shared__ float wave_sums[kBlockThreads / 64];
  const int wave = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) wave_sums[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    float block_sum = 0.f;
#pragma unroll
    for (int w = 0; w < kBlockThreads / 64; ++w) block_sum += wave_sums[w];
    atomicAdd(out, block_sum);  // device-scope by default on CDNA
  }
}

// coef = min(1, max_norm / (sqrt(norm²) + 1e-6)); also writes total_norm
__global__ void clip_coef_kernel(const float* __restrict__ norm_sq, float max_norm,
                                 float* __restrict__ coef, float* __restrict__ total_norm) {
  float n = sqrtf(*norm_sq);
  *total_norm = n;
  float c = max_norm / (n + 1e-6f);
  *coef = c < 1.f ? c : 1.f;
}

// g *= *coef (multi-tensor)
__global__ void multi_tensor_scale_kernel(TensorListMeta meta, const float* __restrict__ coef) {
  const float c = *coef;
  if (c == 1.f) return;
  const int cid = blockIdx.x;
  const int t = find_tensor(meta.chunk_prefix, meta.n_tensors, cid);
  const int64_t chunk_in_tensor = cid - meta.chunk_prefix[t];
  const int64_t numel = meta.numels[t];
  const int64_t lo = chunk_in_tensor * kChunkSize;
  const int64_t hi = min(lo + kChunkSize, numel);
  float* __restrict__ g = reinterpret_cast<float*>(meta.addrs[t]);

  const int64_t tid = threadIdx.x;
  int64_t i = lo + tid * 4;
  const int64_t vec_end = lo + ((hi - lo) & ~int64_t(3));
  for (; i + 3 < vec_end; i += kBlockThreads * 4) {
    float4 gp = *reinterpret_cast<const float4*>(g + i);
    gp.x *= c; gp.y *= c; gp.z *= c; gp.w *= c;
    *reinterpret_cast<float4*>(g + i) = gp;
  }
  for (int64_t j = vec_end + tid; j < hi; j += kBlockThreads) g[j] *= c;
}

// g *= *inv_scale; sets *found_inf = 1 if any non-finite value is seen.
// (the GradScaler's _amp_foreach_non_finite_check_and_unscale_ equivalent,
// reference: SURVEY.md §2.9 N5)
__global__ void unscale_check_kernel(TensorListMeta meta, const float* __restrict__ inv_scale,
                                     float* __restrict__ found_inf) {
  const float s = *inv_scale;
  const int cid = blockIdx.x;
  const int t = find_tensor(meta.chunk_prefix, meta.n_tensors, cid);
  const int64_t chunk_in_tensor = cid - meta.chunk_prefix[t];
  const int64_t numel = meta.numels[t];
  const int64_t lo = chunk_in_tensor * kChunkSize;
  const int64_t hi = min(lo + kChunkSize, numel);
  float* __restrict__ g = reinterpret_cast<float*>(meta.addrs[t]);

  bool bad = false;
  const int64_t tid = threadIdx.x;
  int64_t i = lo + tid * 4;
  const int64_t vec_end = lo + ((hi - lo) & ~int64_t(3));
  for (; i + 3 < vec_end; i += kBlockThreads * 4) {
    float4 gp = *reinterpret_cast<const float4*>(g + i);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float x = (&gp.x)[k] * s;
      bad |= !isfinite(x);
      (&gp.x)[k] = x;
    }
    *reinterpret_cast<float4*>(g + i) = gp;
  }
  for (int64_t j = vec_end + tid; j < hi; j += kBlockThreads) {
    float x = g[j] * s;
    bad |= !isfinite(x);
    g[j] = x;
  }
  if (__any(bad)) {
    if ((threadIdx.x & 63) == 0) *found_inf = 1.f;  // racy-OK flag write
  }
}
