// Python bindings for the gfx950 kernel pack (torch extension, built by
// hipcc directly — no hipify, no CUDA shims).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include <ATen/hip/HIPGeneratorImpl.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <vector>

#include "multi_tensor.h"

// kernels (defined in optim_kernels.hip)
__global__ void fused_adamw_kernel(TensorListMeta, float, float, float, float, float,
                                   float, float, const float*, const float*);
__global__ void adamw_incr_step_kernel(float*);
__global__ void fused_adamw_dev_kernel(TensorListMeta, const float*, const float*,
                                       float, float, float, float, const float*, const float*);
__global__ void fused_adamw_bf16_kernel(TensorListMeta, const float*, const float*,
                                        float, float, float, float, const float*, const float*);
__global__ void l2norm_squared_kernel(TensorListMeta, float*);
__global__ void multi_tensor_copy_kernel(TensorListMeta, bool);
__global__ void clip_coef_kernel(const float*, float, float*, float*);
__global__ void multi_tensor_scale_kernel(TensorListMeta, const float*);
__global__ void unscale_check_kernel(TensorListMeta, const float*, float*);

namespace {

struct MetaHolder {
  at::Tensor addrs_numels;  // int64 device: [n_lists*n + n]
  at::Tensor prefix;        // int32 device: [n+1]
  TensorListMeta meta;
  int total_chunks;
};

MetaHolder build_meta(const std::vector<std::vector<at::Tensor>>& lists) {
  const int n_lists = static_cast<int>(lists.size());
  const int n = static_cast<int>(lists[0].size());
  TORCH_CHECK(n > 0, "empty tensor list");
  auto cpu_i64 = at::empty({n_lists * n + n}, at::TensorOptions().dtype(at::kLong).pinned_memory(true));
  auto cpu_i32 = at::empty({n + 1}, at::TensorOptions().dtype(at::kInt).pinned_memory(true));
  int64_t* a = cpu_i64.data_ptr<int64_t>();
  int32_t* pre = cpu_i32.data_ptr<int32_t>();
  pre[0] = 0;
  for (int t = 0; t < n; ++t) {
    const auto& t0 = lists[0][t];
    TORCH_CHECK(t0.is_cuda() && t0.is_contiguous() && t0.scalar_type() == at::kFloat,
                "multi-tensor ops need contiguous fp32 HIP tensors (tensor ", t, ")");
    int64_t numel = t0.numel();
    a[n_lists * n + t] = numel;
    pre[t + 1] = pre[t] + static_cast<int32_t>((numel + kChunkSize - 1) / kChunkSize);
    for (int l = 0; l < n_lists; ++l) {
      TORCH_CHECK(lists[l][t].numel() == numel, "tensor list shape mismatch");
      a[l * n + t] = reinterpret_cast<int64_t>(lists[l][t].data_ptr());
    }
  }
  MetaHolder h;
  auto dev = lists[0][0].device();
  h.addrs_numels = cpu_i64.to(dev, /*non_blocking=*/true);
  h.prefix = cpu_i32.to(dev, /*non_blocking=*/true);
  h.meta.addrs = h.addrs_numels.data_ptr<int64_t>();
  h.meta.numels = h.addrs_numels.data_ptr<int64_t>() + n_lists * n;
  h.meta.chunk_prefix = h.prefix.data_ptr<int32_t>();
  h.meta.n_tensors = n;
  h.meta.n_lists = n_lists;
  h.total_chunks = pre[n];
  return h;
}

}  // namespace

void fused_adamw(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                 std::vector<at::Tensor> exp_avgs, std::vector<at::Tensor> exp_avg_sqs,
                 int64_t step, double lr, double beta1, double beta2, double eps,
                 double weight_decay,
                 c10::optional<at::Tensor> grad_scale, c10::optional<at::Tensor> found_inf) {
  TORCH_CHECK(params.size() == grads.size() && params.size() == exp_avgs.size() &&
                  params.size() == exp_avg_sqs.size(),
              "fused_adamw: list length mismatch");
  auto h = build_meta({params, grads, exp_avgs, exp_avg_sqs});
  const float bc1 = 1.f - powf(static_cast<float>(beta1), static_cast<float>(step));
  const float bc2 = 1.f - powf(static_cast<float>(beta2), static_cast<float>(step));
  const float rsqrt_bc2 = 1.0f / sqrtf(bc2);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(fused_adamw_kernel, dim3(h.total_chunks), dim3(kBlockThreads), 0, stream.stream(),
                     h.meta, (float)lr, (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
                     bc1, rsqrt_bc2,
                     grad_scale.has_value() ? grad_scale->data_ptr<float>() : nullptr,
                     found_inf.has_value() ? found_inf->data_ptr<float>() : nullptr);
}

// hipGraph-capturable AdamW over a PRE-BUILT device plan. `addrs_numels` is
// the int64 device tensor [4*n ptrs | n numels]; `chunk_prefix` int32 [n+1];
// `step`/`lr` are 1-elem fp32 device tensors. The caller (Python FusedAdamW)
// caches the plan, so a captured graph replays with stable pointers and the
// in-graph step increment keeps bias correction advancing across replays.
void fused_adamw_planned(at::Tensor addrs_numels, at::Tensor chunk_prefix,
                         int64_t n_tensors, int64_t total_chunks,
                         at::Tensor step, at::Tensor lr,
                         double beta1, double beta2, double eps, double weight_decay,
                         c10::optional<at::Tensor> grad_scale, c10::optional<at::Tensor> found_inf,
                         bool bf16_master = false) {
  TensorListMeta meta;
  meta.addrs = addrs_numels.data_ptr<int64_t>();
  meta.numels = addrs_numels.data_ptr<int64_t>() + (bf16_master ? 5 : 4) * n_tensors;
  meta.chunk_prefix = chunk_prefix.data_ptr<int32_t>();
  meta.n_tensors = static_cast<int32_t>(n_tensors);
  meta.n_lists = bf16_master ? 5 : 4;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(adamw_incr_step_kernel, dim3(1), dim3(1), 0, stream.stream(),
                     step.data_ptr<float>());
  if (bf16_master) {
    hipLaunchKernelGGL(fused_adamw_bf16_kernel, dim3(total_chunks), dim3(kBlockThreads), 0, stream.stream(),
                       meta, step.data_ptr<float>(), lr.data_ptr<float>(),
                       (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
                       grad_scale.has_value() ? grad_scale->data_ptr<float>() : nullptr,
                       found_inf.has_value() ? found_inf->data_ptr<float>() : nullptr);
  } else {
    hipLaunchKernelGGL(fused_adamw_dev_kernel, dim3(total_chunks), dim3(kBlockThreads), 0, stream.stream(),
                       meta, step.data_ptr<float>(), lr.data_ptr<float>(),
                       (float)beta1, (float)beta2, (float)eps, (float)weight_decay,
                       grad_scale.has_value() ? grad_scale->data_ptr<float>() : nullptr,
                       found_inf.has_value() ? found_inf->data_ptr<float>() : nullptr);
  }
}

// gather (to_flat=true) or scatter (false) between tensors and flat[offsets]
void multi_tensor_copy(std::vector<at::Tensor> tensors, at::Tensor flat,
                       std::vector<int64_t> offsets, bool to_flat) {
  const int n = (int)tensors.size();
  TORCH_CHECK(n > 0 && (int)offsets.size() == n, "multi_tensor_copy: bad lists");
  const int64_t esize = flat.element_size();
  auto cpu_i64 = at::empty({3 * n}, at::TensorOptions().dtype(at::kLong).pinned_memory(true));
  auto cpu_i32 = at::empty({n + 1}, at::TensorOptions().dtype(at::kInt).pinned_memory(true));
  int64_t* a = cpu_i64.data_ptr<int64_t>();
  int32_t* pre = cpu_i32.data_ptr<int32_t>();
  pre[0] = 0;
  const int64_t flat_base = reinterpret_cast<int64_t>(flat.data_ptr());
  for (int t = 0; t < n; ++t) {
    TORCH_CHECK(tensors[t].is_contiguous() && tensors[t].element_size() == esize,
                "multi_tensor_copy: dtype/layout mismatch at ", t);
    const int64_t nbytes = tensors[t].numel() * esize;
    a[t] = reinterpret_cast<int64_t>(tensors[t].data_ptr());
    a[n + t] = flat_base + offsets[t] * esize;
    a[2 * n + t] = nbytes;
    pre[t + 1] = pre[t] + (int32_t)((nbytes + kChunkSize - 1) / kChunkSize);
  }
  auto dev = flat.device();
  auto dev_i64 = cpu_i64.to(dev, true);
  auto dev_i32 = cpu_i32.to(dev, true);
  TensorListMeta meta;
  meta.addrs = dev_i64.data_ptr<int64_t>();
  meta.numels = dev_i64.data_ptr<int64_t>() + 2 * n;
  meta.chunk_prefix = dev_i32.data_ptr<int32_t>();
  meta.n_tensors = n;
  meta.n_lists = 2;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(multi_tensor_copy_kernel, dim3(pre[n]), dim3(kBlockThreads), 0, stream.stream(),
                     meta, to_flat);
}

// capture-safe variant over a pre-built device plan (addrs: [2n ptrs | n byte
// lengths], prefix over kChunkSize-byte chunks) — the Python caller caches
// the plan (and its pinned staging) exactly like FusedAdamW's plans.
void multi_tensor_copy_planned(at::Tensor addrs_dev, at::Tensor prefix_dev,
                               int64_t n_tensors, int64_t total_chunks, bool to_flat) {
  TensorListMeta meta;
  meta.addrs = addrs_dev.data_ptr<int64_t>();
  meta.numels = addrs_dev.data_ptr<int64_t>() + 2 * n_tensors;
  meta.chunk_prefix = prefix_dev.data_ptr<int32_t>();
  meta.n_tensors = (int32_t)n_tensors;
  meta.n_lists = 2;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(multi_tensor_copy_kernel, dim3(total_chunks), dim3(kBlockThreads), 0, stream.stream(),
                     meta, to_flat);
}

at::Tensor l2norm_squared(std::vector<at::Tensor> grads) {
  auto h = build_meta({grads});
  auto out = at::zeros({1}, grads[0].options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(l2norm_squared_kernel, dim3(h.total_chunks), dim3(kBlockThreads), 0, stream.stream(),
                     h.meta, out.data_ptr<float>());
  return out;
}

// Fully on-device clip_grad_norm_: returns total_norm (0-dim device tensor),
// no host sync anywhere.
at::Tensor clip_grad_norm(std::vector<at::Tensor> grads, double max_norm) {
  auto h = build_meta({grads});
  auto opts = grads[0].options().dtype(at::kFloat);
  auto norm_sq = at::zeros({1}, opts);
  auto coef = at::empty({1}, opts);
  auto total_norm = at::empty({1}, opts);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(l2norm_squared_kernel, dim3(h.total_chunks), dim3(kBlockThreads), 0, stream.stream(),
                     h.meta, norm_sq.data_ptr<float>());
  hipLaunchKernelGGL(clip_coef_kernel, dim3(1), dim3(1), 0, stream.stream(),
                     norm_sq.data_ptr<float>(), (float)max_norm, coef.data_ptr<float>(),
                     total_norm.data_ptr<float>());
  hipLaunchKernelGGL(multi_tensor_scale_kernel, dim3(h.total_chunks), dim3(kBlockThreads), 0, stream.stream(),
                     h.meta, coef.data_ptr<float>());
  return total_norm.squeeze();
}

void multi_tensor_scale(std::vector<at::Tensor> grads, at::Tensor coef) {
  auto h = build_meta({grads});
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(multi_tensor_scale_kernel, dim3(h.total_chunks), dim3(kBlockThreads), 0, stream.stream(),
                     h.meta, coef.data_ptr<float>());
}

void unscale_and_check(std::vector<at::Tensor> grads, at::Tensor inv_scale, at::Tensor found_inf) {
  auto h = build_meta({grads});
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(unscale_check_kernel, dim3(h.total_chunks), dim3(kBlockThreads), 0, stream.stream(),
                     h.meta, inv_scale.data_ptr<float>(), found_inf.data_ptr<float>());
}

// norm kernels (norm_kernels.hip)
__global__ void layernorm_fwd_bf16(const ushort*, const ushort*, const ushort*, ushort*,
                                   float*, float*, int64_t, int, float);
extern "C" hipError_t launch_layernorm_bwd(const void*, const void*, const void*, const void*,
                                           const void*, void*, void*, void*, long long, int, int,
                                           hipStream_t);
__global__ void rmsnorm_fwd_bf16(const ushort*, const ushort*, ushort*, float*, int64_t, int, float);
__global__ void rmsnorm_fwd_wide(const ushort*, const ushort*, ushort*, float*, int64_t, int, float);
extern "C" hipError_t launch_rmsnorm_bwd(const void*, const void*, const void*, const void*,
                                         void*, void*, long long, int, int, hipStream_t);

static const ushort* bfp(const at::Tensor& t) { return reinterpret_cast<const ushort*>(t.data_ptr()); }
static ushort* bfp_mut(at::Tensor& t) { return reinterpret_cast<ushort*>(t.data_ptr()); }

std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor w, at::Tensor b, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.scalar_type() == at::kBFloat16, "layernorm_fwd: bf16 contiguous");
  const int d = (int)x.size(-1);
  TORCH_CHECK(d % 8 == 0, "layernorm_fwd: inner dim must be a multiple of 8");
  const int64_t rows = x.numel() / d;
  auto y = at::empty_like(x);
  auto mean = at::empty({rows}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({rows}, x.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(layernorm_fwd_bf16, dim3(rows), dim3(64), 0, stream.stream(),
                     bfp(x), bfp(w), bfp(b), bfp_mut(y), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     rows, d, (float)eps);
  return {y, mean, rstd};
}


// wave-per-row backward: each block runs 4 rows concurrently. More blocks =
// more waves in flight but more partial-row fold traffic; 512 (2048 waves,
// 2/SIMD) balanced best on MI355X. ACCELERATE_AMD_NORM_BLOCKS overrides.
static int norm_bwd_blocks(int64_t rows) {
  static int cap = []() {
    const char* e = getenv("ACCELERATE_AMD_NORM_BLOCKS");
    return e ? atoi(e) : 512;
  }();
  return (int)std::max<int64_t>(1, std::min<int64_t>((rows + 3) / 4, cap));
}

std::vector<at::Tensor> layernorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                      at::Tensor mean, at::Tensor rstd) {
  const int d = (int)x.size(-1);
  TORCH_CHECK(d <= 2048, "layernorm_bwd: fused path supports inner dim <= 2048");
  const int64_t rows = x.numel() / d;
  const int n_blocks = norm_bwd_blocks(rows);
  const int n_partials = n_blocks * 4;  // one partial row per wave
  auto dx = at::empty_like(x);
  // dw and db interleave in ONE [P][2d] buffer -> one fold launch
  auto dwdb_partial = at::empty({n_partials, 2 * (int64_t)d}, x.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  auto dyc = dy.contiguous();
  auto err = launch_layernorm_bwd(bfp(dyc), bfp(x), bfp(w), mean.data_ptr<float>(),
                                  rstd.data_ptr<float>(), bfp_mut(dx), dwdb_partial.data_ptr<float>(),
                                  nullptr, rows, d, n_blocks, stream.stream());
  TORCH_CHECK(err == hipSuccess, "layernorm_bwd: ", hipGetErrorString(err));
  // fold with torch's tree reduction (a serial-loop fold kernel was the old
  // bottleneck: d/256 blocks looping n_partials rows, latency-bound)
  auto folded = dwdb_partial.sum(0).to(x.scalar_type());
  return {dx, folded.narrow(0, 0, d), folded.narrow(0, d, d)};
}

std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.scalar_type() == at::kBFloat16, "rmsnorm_fwd: bf16 contiguous");
  const int d = (int)x.size(-1);
  TORCH_CHECK(d % 8 == 0, "rmsnorm_fwd: inner dim must be a multiple of 8");
  const int64_t rows = x.numel() / d;
  auto y = at::empty_like(x);
  auto rstd = at::empty({rows}, x.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  if (rows < 4)  // decode shapes: 256 threads cooperate per row
    hipLaunchKernelGGL(rmsnorm_fwd_wide, dim3(rows), dim3(256), 0, stream.stream(),
                       bfp(x), bfp(w), bfp_mut(y), rstd.data_ptr<float>(), rows, d, (float)eps);
  else
    hipLaunchKernelGGL(rmsnorm_fwd_bf16, dim3(rows), dim3(64), 0, stream.stream(),
                       bfp(x), bfp(w), bfp_mut(y), rstd.data_ptr<float>(), rows, d, (float)eps);
  return {y, rstd};
}

std::vector<at::Tensor> rmsnorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w, at::Tensor rstd) {
  const int d = (int)x.size(-1);
  TORCH_CHECK(d <= 8192, "rmsnorm_bwd: fused path supports inner dim <= 8192");
  const int64_t rows = x.numel() / d;
  const int n_blocks = norm_bwd_blocks(rows);
  const int n_partials = n_blocks * 4;
  auto dx = at::empty_like(x);
  auto dw_partial = at::empty({n_partials, d}, x.options().dtype(at::kFloat));
  auto dw = at::empty({d}, x.options());
  auto stream = at::hip::getCurrentHIPStream();
  auto dyc = dy.contiguous();
  auto err = launch_rmsnorm_bwd(bfp(dyc), bfp(x), bfp(w), rstd.data_ptr<float>(), bfp_mut(dx),
                                dw_partial.data_ptr<float>(), rows, d, n_blocks, stream.stream());
  TORCH_CHECK(err == hipSuccess, "rmsnorm_bwd: ", hipGetErrorString(err));
  dw.copy_(dw_partial.sum(0));
  return {dx, dw};
}

// MFMA GEMM (gemm_kernels.hip 128^2 / gemm8_kernels.hip 256^2 8-phase)
__global__ void gemm_bt_bf16_kernel(const __bf16*, const __bf16*, const ushort*, ushort*, int, int, int);
hipError_t launch_gemm8_bt(const void*, const void*, const void*, void*, int, int, int, hipStream_t);

at::Tensor mfma_gemm_bt(at::Tensor a, at::Tensor b, c10::optional<at::Tensor> bias) {
  // C[M,N] = a[M,K] @ b[N,K]^T — the nn.Linear forward layout
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && a.scalar_type() == at::kBFloat16, "mfma_gemm_bt: A bf16 contiguous");
  TORCH_CHECK(b.is_cuda() && b.is_contiguous() && b.scalar_type() == at::kBFloat16, "mfma_gemm_bt: B bf16 contiguous");
  const int M = (int)a.size(0), K = (int)a.size(1), N = (int)b.size(0);
  TORCH_CHECK(b.size(1) == K, "mfma_gemm_bt: K mismatch");
  TORCH_CHECK(M % 128 == 0 && N % 128 == 0 && K % 64 == 0, "mfma_gemm_bt: need M,N %128==0 and K %64==0");
  auto c = at::empty({M, N}, a.options());
  auto stream = at::hip::getCurrentHIPStream();
  if (M % 256 == 0 && N % 256 == 0 && K % 128 == 0) {
    // the 8-phase 256^2 template (1079 TF/s @4096^3 vs 652 for the 128^2)
    hipError_t e = launch_gemm8_bt(a.data_ptr(), b.data_ptr(),
                                   bias.has_value() ? bias->data_ptr() : nullptr,
                                   c.data_ptr(), M, N, K, stream.stream());
    TORCH_CHECK(e == hipSuccess, "gemm8_bt launch failed: ", hipGetErrorString(e));
    return c;
  }
  const int grid = (M / 128) * (N / 128);
  hipLaunchKernelGGL(gemm_bt_bf16_kernel, dim3(grid), dim3(256), 0, stream.stream(),
                     reinterpret_cast<const __bf16*>(a.data_ptr()),
                     reinterpret_cast<const __bf16*>(b.data_ptr()),
                     bias.has_value() ? reinterpret_cast<const ushort*>(bias->data_ptr()) : nullptr,
                     reinterpret_cast<ushort*>(c.data_ptr()), M, N, K);
  return c;
}

// rope kernel (rope_kernels.hip)
__global__ void rope_bf16_kernel(const ushort*, ushort*, const float*, const float*,
                                 int64_t, int, int, float);

at::Tensor rope_bf16(at::Tensor x, at::Tensor cos_t, at::Tensor sin_t, bool inverse) {
  // x: [..., S, D] bf16 contiguous; cos/sin: [>=S, D/2] fp32 contiguous
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.scalar_type() == at::kBFloat16, "rope: bf16 contiguous");
  const int D = (int)x.size(-1);
  const int S = (int)x.size(-2);
  TORCH_CHECK(D % 16 == 0, "rope: head_dim must be a multiple of 16");
  TORCH_CHECK(cos_t.size(-1) == D / 2 && cos_t.size(0) >= S, "rope: bad cos table");
  const int64_t n_bh = x.numel() / ((int64_t)S * D);
  auto y = at::empty_like(x);
  const int64_t total = n_bh * S * (D / 16);
  int grid = (int)std::min<int64_t>((total + 255) / 256, 2048);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(rope_bf16_kernel, dim3(grid), dim3(256), 0, stream.stream(),
                     reinterpret_cast<const ushort*>(x.data_ptr()),
                     reinterpret_cast<ushort*>(y.data_ptr()),
                     cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
                     n_bh, S, D, inverse ? -1.f : 1.f);
  return y;
}

// fp8 kernels (fp8_kernels.hip)
__global__ void fp8_cast_amax_e4m3(const ushort*, unsigned char*, const float*, float*, int64_t);
__global__ void fp8_cast_amax_e5m2(const ushort*, unsigned char*, const float*, float*, int64_t);
__global__ void fp8_update_scale(const float*, int, float, float, float*, float*);
__global__ void fp8_cast_transpose_e4m3(const ushort*, unsigned char*, unsigned char*,
                                        const float*, float*, int64_t, int64_t);
__global__ void fp8_cast_transpose_e5m2(const ushort*, unsigned char*, unsigned char*,
                                        const float*, float*, int64_t, int64_t);

void fp8_cast_amax(at::Tensor in, at::Tensor out, at::Tensor scale, at::Tensor amax, bool e5m2) {
  TORCH_CHECK(in.is_cuda() && in.is_contiguous() && in.scalar_type() == at::kBFloat16,
              "fp8_cast_amax: input must be contiguous bf16 on GPU");
  TORCH_CHECK(out.numel() == in.numel() && out.element_size() == 1, "fp8_cast_amax: bad output");
  int64_t n = in.numel();
  // memory-bound: cap grid and grid-stride (CDNA4 Guideline 11); 16 elems/thread
  int grid = (int)std::min<int64_t>((n + kBlockThreads * 16 - 1) / (kBlockThreads * 16), 2048);
  auto stream = at::hip::getCurrentHIPStream();
  auto* inp = reinterpret_cast<const ushort*>(in.data_ptr());
  auto* outp = reinterpret_cast<unsigned char*>(out.data_ptr());
  if (e5m2) {
    hipLaunchKernelGGL(fp8_cast_amax_e5m2, dim3(grid), dim3(kBlockThreads), 0, stream.stream(),
                       inp, outp, scale.data_ptr<float>(), amax.data_ptr<float>(), n);
  } else {
    hipLaunchKernelGGL(fp8_cast_amax_e4m3, dim3(grid), dim3(kBlockThreads), 0, stream.stream(),
                       inp, outp, scale.data_ptr<float>(), amax.data_ptr<float>(), n);
  }
}

void fp8_cast_transpose(at::Tensor in, at::Tensor out, at::Tensor out_t,
                        at::Tensor scale, at::Tensor amax, bool e5m2) {
  TORCH_CHECK(in.is_cuda() && in.is_contiguous() && in.scalar_type() == at::kBFloat16 && in.dim() == 2,
              "fp8_cast_transpose: 2-D contiguous bf16 input required");
  const int64_t R = in.size(0), C = in.size(1);
  TORCH_CHECK(R % 128 == 0 && C % 128 == 0, "fp8_cast_transpose: dims must be multiples of 128");
  TORCH_CHECK(out.numel() == R * C && out_t.numel() == R * C, "fp8_cast_transpose: bad outputs");
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(C / 128, R / 128);
  auto* inp = reinterpret_cast<const ushort*>(in.data_ptr());
  auto* o = reinterpret_cast<unsigned char*>(out.data_ptr());
  auto* ot = reinterpret_cast<unsigned char*>(out_t.data_ptr());
  if (e5m2) {
    hipLaunchKernelGGL(fp8_cast_transpose_e5m2, grid, dim3(256), 0, stream.stream(),
                       inp, o, ot, scale.data_ptr<float>(), amax.data_ptr<float>(), R, C);
  } else {
    hipLaunchKernelGGL(fp8_cast_transpose_e4m3, grid, dim3(256), 0, stream.stream(),
                       inp, o, ot, scale.data_ptr<float>(), amax.data_ptr<float>(), R, C);
  }
}

void fp8_update_scale_fn(at::Tensor history, double fp8_max, double margin_pow2,
                         at::Tensor scale, at::Tensor scale_inv) {
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(fp8_update_scale, dim3(1), dim3(1), 0, stream.stream(),
                     history.data_ptr<float>(), (int)history.numel(), (float)fp8_max,
                     (float)margin_pow2, scale.data_ptr<float>(), scale_inv.data_ptr<float>());
}

// fused flash-attention forward (attention_kernels.hip)
#include "attention.h"
hipError_t launch_fa_fwd(const void*, const void*, const void*, void*, float*,
                         int64_t, int, int, int, int, int, float, int, int,
                         const Str3*, hipStream_t);

namespace {

// (batch, head, seq) element strides of a [B,H,S,D] view; dim 3 must be
// contiguous and every access 16-byte aligned (strides % 8 elements).
Str3 str3_of(const at::Tensor& t, const char* who) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kBFloat16 && t.dim() == 4,
              who, ": 4-D bf16 HIP tensor required");
  TORCH_CHECK(t.stride(3) == 1, who, ": innermost dim must be contiguous");
  Str3 s{t.stride(0), t.stride(1), t.stride(2)};
  TORCH_CHECK(s.b % 8 == 0 && s.h % 8 == 0 && s.s % 8 == 0,
              who, ": strides must be multiples of 8 elements (16 B)");
  return s;
}

}  // namespace

std::vector<at::Tensor> flash_attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v,
                                       bool causal, double scale, int64_t past) {
  // q:[B,Hq,Sq,D], k/v:[B,Hkv,Sk,D] — ANY strides with dim 3 contiguous
  // (transposed BSHD views and un-expanded GQA kv caches read zero-copy).
  const int64_t B = q.size(0), Hq = q.size(1), Sq = q.size(2), D = q.size(3);
  const int64_t Hkv = k.size(1), Sk = k.size(2);
  TORCH_CHECK(k.size(0) == B && k.size(3) == D && v.sizes() == k.sizes(),
              "flash_attn_fwd: k/v shape mismatch");
  TORCH_CHECK(Hq % Hkv == 0, "flash_attn_fwd: Hq must be a multiple of Hkv");
  TORCH_CHECK(D == 64 || D == 128, "flash_attn_fwd: head_dim must be 64 or 128");
  TORCH_CHECK(B * Hq <= 65535, "flash_attn_fwd: grid.y overflow");
  // output in BSHD storage: the model's .transpose(1,2).reshape(B,S,H*D)
  // after attention becomes a free view
  auto out = at::empty({B, Sq, Hq, D}, q.options()).permute({0, 2, 1, 3});
  auto lse = at::empty({B, Hq, Sq}, q.options().dtype(at::kFloat));
  Str3 strides[4] = {str3_of(q, "q"), str3_of(k, "k"), str3_of(v, "v"), str3_of(out, "out")};
  auto stream = at::hip::getCurrentHIPStream();
  hipError_t e = launch_fa_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), out.data_ptr(),
                               lse.data_ptr<float>(), B * Hq, (int)Sq, (int)Sk, (int)D,
                               (int)past, causal ? 1 : 0, (float)scale, (int)Hq, (int)Hkv,
                               strides, stream.stream());
  TORCH_CHECK(e == hipSuccess, "flash_attn_fwd launch failed: ", hipGetErrorString(e));
  return {out, lse};
}

hipError_t launch_fa_bwd(const void*, const void*, const void*, const void*, const void*,
                         const float*, float*, void*, void*, void*, int64_t, int, int, int,
                         int, int, float, int, int, const Str3*, hipStream_t);

std::vector<at::Tensor> flash_attn_bwd(at::Tensor dout, at::Tensor q, at::Tensor k,
                                       at::Tensor v, at::Tensor out, at::Tensor lse,
                                       bool causal, double scale, int64_t past) {
  // Returns (dq [B,Hq,Sq,D] contiguous, dk/dv PARTIALS [B,Hq,Sk,D]
  // contiguous — the Python wrapper sums the GQA groups down to Hkv).
  const int64_t B = q.size(0), Hq = q.size(1), Sq = q.size(2), D = q.size(3);
  const int64_t Hkv = k.size(1), Sk = k.size(2);
  TORCH_CHECK(D == 64 || D == 128, "flash_attn_bwd: head_dim must be 64 or 128");
  TORCH_CHECK(Hq % Hkv == 0, "flash_attn_bwd: Hq must be a multiple of Hkv");
  if (dout.stride(3) != 1) dout = dout.contiguous();
  auto dq = at::empty({B, Hq, Sq, D}, q.options());
  auto dk = at::empty({B, Hq, Sk, D}, q.options());
  auto dv = at::empty({B, Hq, Sk, D}, q.options());
  auto drow = at::empty({B, Hq, Sq}, q.options().dtype(at::kFloat));
  Str3 strides[5] = {str3_of(q, "q"), str3_of(k, "k"), str3_of(v, "v"),
                     str3_of(dout, "dout"), str3_of(out, "out")};
  auto stream = at::hip::getCurrentHIPStream();
  hipError_t e = launch_fa_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), dout.data_ptr(),
                               out.data_ptr(), lse.data_ptr<float>(), drow.data_ptr<float>(),
                               dq.data_ptr(), dk.data_ptr(), dv.data_ptr(), B * Hq,
                               (int)Sq, (int)Sk, (int)D, (int)past, causal ? 1 : 0,
                               (float)scale, (int)Hq, (int)Hkv, strides, stream.stream());
  TORCH_CHECK(e == hipSuccess, "flash_attn_bwd launch failed: ", hipGetErrorString(e));
  return {dq, dk, dv};
}

// weight-only quantization kernels (quant_kernels.hip)
__global__ void int8_dequant_kernel(const char*, const float*, ushort*, int64_t, int64_t);
__global__ void int4_dequant_kernel(const unsigned char*, const float*, ushort*, int64_t, int64_t, int);
__global__ void w8a16_gemv_kernel(const char*, const float*, const ushort*, const ushort*,
                                  ushort*, int64_t, int64_t, int);

at::Tensor int8_dequant(at::Tensor q, at::Tensor scale) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && q.scalar_type() == at::kChar && q.dim() == 2,
              "int8_dequant: 2-D contiguous int8 on GPU");
  const int64_t rows = q.size(0), cols = q.size(1);
  TORCH_CHECK(cols % 16 == 0, "int8_dequant: cols must be a multiple of 16");
  TORCH_CHECK(scale.numel() == rows && scale.scalar_type() == at::kFloat, "int8_dequant: bad scale");
  auto out = at::empty({rows, cols}, q.options().dtype(at::kBFloat16));
  int grid = (int)std::min<int64_t>((rows * cols / 16 + kBlockThreads - 1) / kBlockThreads, 2048);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(int8_dequant_kernel, dim3(grid), dim3(kBlockThreads), 0, stream.stream(),
                     reinterpret_cast<const char*>(q.data_ptr()), scale.data_ptr<float>(),
                     reinterpret_cast<ushort*>(out.data_ptr()), rows, cols);
  return out;
}

at::Tensor int4_dequant(at::Tensor q, at::Tensor scale, int64_t cols, int64_t group) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && q.scalar_type() == at::kByte && q.dim() == 2,
              "int4_dequant: 2-D contiguous uint8 (packed nibbles) on GPU");
  const int64_t rows = q.size(0);
  TORCH_CHECK(q.size(1) * 2 == cols, "int4_dequant: packed width mismatch");
  TORCH_CHECK(cols % 16 == 0 && group % 16 == 0 && cols % group == 0, "int4_dequant: bad group");
  TORCH_CHECK(scale.numel() == rows * (cols / group) && scale.scalar_type() == at::kFloat,
              "int4_dequant: bad scale");
  auto out = at::empty({rows, cols}, q.options().dtype(at::kBFloat16));
  int grid = (int)std::min<int64_t>((rows * cols / 16 + kBlockThreads - 1) / kBlockThreads, 2048);
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(int4_dequant_kernel, dim3(grid), dim3(kBlockThreads), 0, stream.stream(),
                     reinterpret_cast<const unsigned char*>(q.data_ptr()), scale.data_ptr<float>(),
                     reinterpret_cast<ushort*>(out.data_ptr()), rows, cols, (int)group);
  return out;
}

at::Tensor w8a16_gemv(at::Tensor q, at::Tensor scale, at::Tensor x, c10::optional<at::Tensor> bias) {
  // y[b, rows] = x[b, cols] @ q^T * scale (decode-shaped: batch <= 8)
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && q.scalar_type() == at::kChar, "w8a16_gemv: int8 weight");
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.scalar_type() == at::kBFloat16, "w8a16_gemv: bf16 x");
  const int64_t rows = q.size(0), cols = q.size(1);
  TORCH_CHECK(x.size(-1) == cols && cols % 1024 == 0, "w8a16_gemv: cols must be a multiple of 1024");
  const int batch = (int)(x.numel() / cols);
  TORCH_CHECK(batch <= 8, "w8a16_gemv: decode path only (batch <= 8)");
  auto sizes = x.sizes().vec();
  sizes.back() = rows;
  auto y = at::empty(sizes, x.options());
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(w8a16_gemv_kernel, dim3((rows + 3) / 4), dim3(256), 0, stream.stream(),
                     reinterpret_cast<const char*>(q.data_ptr()), scale.data_ptr<float>(),
                     reinterpret_cast<const ushort*>(x.data_ptr()),
                     bias.has_value() ? reinterpret_cast<const ushort*>(bias->data_ptr()) : nullptr,
                     reinterpret_cast<ushort*>(y.data_ptr()), rows, cols, batch);
  return y;
}

extern "C" hipError_t launch_gemv_bf16(const void*, const void*, const void*, void*,
                                       long long, long long, int, hipStream_t);

at::Tensor gemv_bf16(at::Tensor x, at::Tensor w, c10::optional<at::Tensor> bias) {
  // y = x @ w^T for skinny x (decode): x [M, K] (M <= 8), w [N, K]
  TORCH_CHECK(x.is_cuda() && x.scalar_type() == at::kBFloat16 && w.scalar_type() == at::kBFloat16,
              "gemv_bf16: bf16 cuda");
  auto xc = x.contiguous();
  TORCH_CHECK(w.is_contiguous(), "gemv_bf16: weight must be contiguous");
  const int64_t K = w.size(1), N = w.size(0);
  const int64_t M = xc.numel() / K;
  TORCH_CHECK(M >= 1 && M <= 8 && xc.size(-1) == K && K % 8 == 0, "gemv_bf16: shape");
  auto y = at::empty({M, N}, x.options());
  auto stream = at::hip::getCurrentHIPStream();
  auto err = launch_gemv_bf16(w.data_ptr(), xc.data_ptr(),
                              bias.has_value() ? bias->data_ptr() : nullptr, y.data_ptr(),
                              N, K, (int)M, stream.stream());
  TORCH_CHECK(err == hipSuccess, "gemv_bf16: ", hipGetErrorString(err));
  std::vector<int64_t> shape(x.sizes().begin(), x.sizes().end());
  shape.back() = N;
  return y.view(shape);
}

extern "C" hipError_t launch_dropout_add_ln_fwd(const void*, const void*, const void*,
                                                const void*, void*, void*, void*, void*, void*,
                                                long long, int, float, float, float,
                                                at::PhiloxCudaState, int, hipStream_t);
extern "C" hipError_t launch_dropout_add_ln_bwd(const void*, const void*, const void*,
                                                const void*, const void*, const void*, void*,
                                                void*, void*, long long, int, float, int,
                                                hipStream_t);

std::vector<at::Tensor> dropout_add_ln_fwd(at::Tensor x, at::Tensor z, at::Tensor w,
                                           at::Tensor b, double eps, double p) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.scalar_type() == at::kBFloat16,
              "dropout_add_ln_fwd: bf16 contiguous");
  const int d = (int)x.size(-1);
  TORCH_CHECK(d % 8 == 0 && d <= 2048, "dropout_add_ln_fwd: inner dim");
  const int64_t rows = x.numel() / d;
  auto y = at::empty_like(x);
  auto s_out = at::empty_like(x);
  auto mask = at::empty({rows, d}, x.options().dtype(at::kByte));
  auto mean = at::empty({rows}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({rows}, x.options().dtype(at::kFloat));
  const float keep = 1.f - (float)p;
  at::PhiloxCudaState rng(0, 0);
  if (keep < 1.f) {
    auto gen = at::get_generator_or_default<at::CUDAGeneratorImpl>(
        std::nullopt, at::cuda::detail::getDefaultCUDAGenerator());
    std::lock_guard<std::mutex> lock(gen->mutex_);
    rng = gen->philox_cuda_state((uint64_t)x.numel());
  }
  const int n_blocks = norm_bwd_blocks(rows);
  auto stream = at::hip::getCurrentHIPStream();
  auto err = launch_dropout_add_ln_fwd(bfp(x), bfp(z), bfp(w), bfp(b), bfp_mut(y),
                                       bfp_mut(s_out), mask.data_ptr(), mean.data_ptr<float>(),
                                       rstd.data_ptr<float>(), rows, d, (float)eps, keep,
                                       keep > 0.f ? 1.f / keep : 0.f, rng, n_blocks,
                                       stream.stream());
  TORCH_CHECK(err == hipSuccess, "dropout_add_ln_fwd: ", hipGetErrorString(err));
  return {y, s_out, mask, mean, rstd};
}

std::vector<at::Tensor> dropout_add_ln_bwd(at::Tensor dy, at::Tensor s, at::Tensor w,
                                           at::Tensor mask, at::Tensor mean, at::Tensor rstd,
                                           double p) {
  const int d = (int)s.size(-1);
  const int64_t rows = s.numel() / d;
  const int n_blocks = norm_bwd_blocks(rows);
  const int n_partials = n_blocks * 4;
  auto dx = at::empty_like(s);
  auto dz = at::empty_like(s);
  auto dwdb_partial = at::empty({n_partials, 2 * (int64_t)d}, s.options().dtype(at::kFloat));
  const float keep = 1.f - (float)p;
  auto stream = at::hip::getCurrentHIPStream();
  auto dyc = dy.contiguous();
  auto err = launch_dropout_add_ln_bwd(bfp(dyc), bfp(s), bfp(w), mask.data_ptr(),
                                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                                       bfp_mut(dx), bfp_mut(dz), dwdb_partial.data_ptr<float>(),
                                       rows, d, keep > 0.f ? 1.f / keep : 0.f, n_blocks,
                                       stream.stream());
  TORCH_CHECK(err == hipSuccess, "dropout_add_ln_bwd: ", hipGetErrorString(err));
  auto folded = dwdb_partial.sum(0).to(s.scalar_type());
  return {dx, dz, folded.narrow(0, 0, d), folded.narrow(0, d, d)};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("layernorm_fwd", &layernorm_fwd, "fused bf16 LayerNorm forward");
  m.def("layernorm_bwd", &layernorm_bwd, "fused bf16 LayerNorm backward");
  m.def("dropout_add_ln_fwd", &dropout_add_ln_fwd,
        "fused y = LayerNorm(x + dropout(z)) forward (graph-safe philox mask)");
  m.def("dropout_add_ln_bwd", &dropout_add_ln_bwd,
        "fused dropout+add+LayerNorm backward -> (dx, dz, dw, db)");
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "fused bf16 RMSNorm forward");
  m.def("rope_bf16", &rope_bf16, "fused rotary embedding (bf16, half-split layout)");
  m.def("mfma_gemm_bt", &mfma_gemm_bt, "hand-written MFMA bf16 GEMM (A @ B^T + bias)");
  m.def("gemv_bf16", &gemv_bf16, "fused bf16 decode GEMV (x @ W^T, M <= 8)",
        py::arg("x"), py::arg("w"), py::arg("bias") = py::none());
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "fused bf16 RMSNorm backward");
  m.def("fp8_cast_amax", &fp8_cast_amax, "bf16 -> fp8 cast with fused amax (gfx950)");
  m.def("fp8_cast_transpose", &fp8_cast_transpose,
        "bf16 -> fp8 + fp8^T with fused amax (LDS-tiled, gfx950)");
  m.def("fp8_update_scale", &fp8_update_scale_fn, "delayed-scaling scale update");
  m.def("flash_attn_fwd", &flash_attn_fwd,
        "fused flash-attention forward (bf16, head_dim 64/128) -> (out, lse)");
  m.def("flash_attn_bwd", &flash_attn_bwd,
        "fused flash-attention backward (logsumexp recompute) -> (dq, dk, dv)");
  m.def("int8_dequant", &int8_dequant, "int8 weight -> bf16 (per-channel scale)");
  m.def("int4_dequant", &int4_dequant, "packed int4 weight -> bf16 (group-wise scale)");
  m.def("w8a16_gemv", &w8a16_gemv, "fused int8-weight x bf16-activation matvec (decode)");
  m.def("fused_adamw", &fused_adamw, "fused multi-tensor AdamW (gfx950)");
  m.def("fused_adamw_planned", &fused_adamw_planned,
        "graph-capturable fused AdamW over a cached device plan",
        pybind11::arg("addrs_numels"), pybind11::arg("chunk_prefix"), pybind11::arg("n_tensors"),
        pybind11::arg("total_chunks"), pybind11::arg("step"), pybind11::arg("lr"),
        pybind11::arg("beta1"), pybind11::arg("beta2"), pybind11::arg("eps"),
        pybind11::arg("weight_decay"), pybind11::arg("grad_scale") = pybind11::none(),
        pybind11::arg("found_inf") = pybind11::none(), pybind11::arg("bf16_master") = false);
  m.def("l2norm_squared", &l2norm_squared, "global L2 norm squared over tensor list");
  m.def("clip_grad_norm", &clip_grad_norm, "on-device clip_grad_norm_, returns total norm");
  m.def("multi_tensor_scale", &multi_tensor_scale, "g *= *coef");
  m.def("multi_tensor_copy", &multi_tensor_copy,
        "fused gather/scatter between tensor list and a flat bucket");
  m.def("multi_tensor_copy_planned", &multi_tensor_copy_planned,
        "capture-safe fused gather/scatter over a cached plan");
  m.def("unscale_and_check", &unscale_and_check, "g *= *inv_scale with non-finite detection");
}
