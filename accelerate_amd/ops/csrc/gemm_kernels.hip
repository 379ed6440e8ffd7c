// Hand-written MFMA GEMM for gfx950: C[M,N] = A[M,K] · B[N,K]^T (+bias),
// bf16 inputs, fp32 accumulate, bf16 out — the nn.Linear forward layout.
//
// Structure = the CDNA4 guide's verified 128²-tile recipe (§5 ladder step 3):
// 4 waves per block (2×2), each wave owns a 64×64 output sub-tile of 4×4
// 16×16 MFMA fragments; BK=64 staged through LDS with
// __builtin_amdgcn_global_load_lds width 16 (direct HBM→LDS, no VGPR
// round-trip); single LDS buffer, two barriers per K-step.
//
// Fragment mapping (verified on MI355X by benchmarks/mfma_layout_probe.hip):
// the mfma_f32_16x16x32_bf16 k-pairing is consistent for any shared (lane,j)
// →k bijection; we use k = (lane>>4)*8 + j so LDS fragment reads are single
// contiguous 16-byte ds_read_b128s. C/D: col = lane&15, row = (lane>>4)*4+j
// (guide §3, m89-verified).

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

typedef __bf16 gbf16;
typedef gbf16 bf16x8g __attribute__((ext_vector_type(8)));
typedef float f32x4g __attribute__((ext_vector_type(4)));

__device__ __forceinline__ ushort gf2bf(float f) {
  union { float f; unsigned int i; } cv;
  cv.f = f;
  if ((cv.i & 0x7F800000u) == 0x7F800000u) return (ushort)(cv.i >> 16) | (ushort)((cv.i & 0xFFFFu) ? 0x40 : 0);
  cv.i += 0x7FFFu + ((cv.i >> 16) & 1u);
  return (ushort)(cv.i >> 16);
}

}  // namespace

// block = 256 threads (4 waves as 2×2), tile 128(M)×128(N), BK=64.
// Double-buffered K-loop with counted prefetch (the CDNA4 guide's minimum
// 2-phase recipe, §5.5 T3): next tile's global_load_lds issues BEFORE the
// current tile's ds_read+MFMA, one vmcnt(0)+barrier per tile.
__global__ __launch_bounds__(256) void gemm_bt_bf16_kernel(
    const gbf16* __restrict__ A, const gbf16* __restrict__ B,
    const ushort* __restrict__ bias, ushort* __restrict__ C,
    int M, int N, int K) {
  constexpr int BM = 128, BN = 128, BK = 64;
  __shared__ gbf16 As[2][BM * BK];
  __shared__ gbf16 Bs[2][BN * BK];

  const int tid = threadIdx.x;
  const int wave = tid >> 6;          // 0..3
  const int lane = tid & 63;
  const int wm = wave >> 1;           // wave row (0..1) → 64 rows each
  const int wn = wave & 1;            // wave col (0..1) → 64 cols each

  // XCD-aware block swizzle (guide T1, bijective variant): consecutive
  // blocks share B-panels; group them per XCD for L2 locality.
  int nwg = gridDim.x;
  int bid = blockIdx.x;
  {
    const int nx = 8;
    const int q = nwg / nx, r = nwg % nx;
    const int xcd = bid % nx, idx = bid / nx;
    bid = (xcd < r ? xcd * (q + 1) : r * (q + 1) + (xcd - r) * q) + idx;
  }
  const int tiles_n = N / BN;
  const int tile_m = (bid / tiles_n) * BM;
  const int tile_n = (bid % tiles_n) * BN;

  // staging: per round, each wave DMAs 8 rows × 64 cols (8 lanes/row × 16 B)
  // into a contiguous LDS span (global_load_lds is wave-uniform-base+lane*16)
  const int s_row_in_wave = lane >> 3;   // 0..7
  const int s_col = (lane & 7) * 8;      // 0..56, 8 bf16 = 16 B

  f32x4g acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = f32x4g{};

  // LDS write address is HARDWARE-fixed at (wave-uniform base + lane*16 B)
  // (guide §5 caveat): with lane = rr*8+cc that lands exactly row-major
  // [8 rows][64 cols] from the base — the global address carries the
  // per-lane row/col.
  auto stage = [&](int buf, int k0) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = r * 32 + wave * 8 + s_row_in_wave;  // 0..127
      const gbf16* ga = A + (int64_t)(tile_m + row) * K + k0 + s_col;
      const gbf16* gb = B + (int64_t)(tile_n + row) * K + k0 + s_col;
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)ga,
          (__attribute__((address_space(3))) unsigned int*)(&As[buf][(r * 32 + wave * 8) * BK]), 16, 0, 0);
      __builtin_amdgcn_global_load_lds(
          (const __attribute__((address_space(1))) unsigned int*)gb,
          (__attribute__((address_space(3))) unsigned int*)(&Bs[buf][(r * 32 + wave * 8) * BK]), 16, 0, 0);
    }
  };

  auto compute = [&](int buf) {
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // two K=32 sub-steps
      const int kslot = kk * 32 + (lane >> 4) * 8;
      bf16x8g af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        af[i] = *reinterpret_cast<const bf16x8g*>(&As[buf][(wm * 64 + i * 16 + (lane & 15)) * BK + kslot]);
        bf[i] = *reinterpret_cast<const bf16x8g*>(&Bs[buf][(wn * 64 + i * 16 + (lane & 15)) * BK + kslot]);
      }
      asm volatile("s_waitcnt lgkmcnt(0)");
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int mi = 0; mi < 4; ++mi)
#pragma unroll
        for (int ni = 0; ni < 4; ++ni)
          acc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[mi], bf[ni], acc[mi][ni], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
  };

  const int nt = K / BK;
  int cur = 0;
  // prologue: stage tile 0 and drain it
  stage(0, 0);
  asm volatile("s_waitcnt vmcnt(0)");
  __syncthreads();
  for (int t = 0; t < nt - 1; ++t) {
    stage(cur ^ 1, (t + 1) * BK);  // issue next tile FIRST (stays in flight)
    compute(cur);
    asm volatile("s_waitcnt vmcnt(0)");
    __syncthreads();
    cur ^= 1;
  }
  compute(cur);  // epilogue tile (no prefetch)

  // epilogue: C[row][col], row = (lane>>4)*4 + j within fragment
#pragma unroll
  for (int mi = 0; mi < 4; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int col = tile_n + wn * 64 + ni * 16 + (lane & 15);
      const float badd = (bias != nullptr) ? (float)(*reinterpret_cast<const __hip_bfloat16*>(bias + col)) : 0.f;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int row = tile_m + wm * 64 + mi * 16 + (lane >> 4) * 4 + j;
        C[(int64_t)row * N + col] = gf2bf(acc[mi][ni][j] + badd);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Decode GEMV: y[m, n] = x[m, :] . W[n, :]  (nn.Linear weight layout),
// bf16 in/out, fp32 accumulate via V_DOT2_F32_BF16 (one VALU per bf16
// pair keeps the loop HBM-bound, not conversion-bound — the reason the
// rocBLAS skinny-GEMM path measured only ~3.4 TB/s on 70B decode).
// One wave per output row, grid-stride; x rows (M <= 8) stay L1-resident.
// ---------------------------------------------------------------------------

namespace {
typedef unsigned short gus16x2 __attribute__((ext_vector_type(2)));
typedef unsigned short gus16x8 __attribute__((ext_vector_type(8)));

__device__ __forceinline__ ushort gvf2bf(float f) {
  union { float f; unsigned i; } cv;
  cv.f = f;
  if ((cv.i & 0x7F800000u) == 0x7F800000u)
    return (ushort)(cv.i >> 16) | (ushort)((cv.i & 0xFFFFu) ? 0x40 : 0);
  cv.i += 0x7FFFu + ((cv.i >> 16) & 1u);
  return (ushort)(cv.i >> 16);
}
}  // namespace

template <int M>
__global__ __launch_bounds__(256) void gemv_bf16_kernel(
    const ushort* __restrict__ w, const ushort* __restrict__ x,
    const ushort* __restrict__ bias, ushort* __restrict__ y,
    int64_t N, int64_t K) {
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  for (int64_t row = (int64_t)blockIdx.x * 4 + wave; row < N; row += (int64_t)gridDim.x * 4) {
    const ushort* wr = w + row * K;
    float acc[M];
#pragma unroll
    for (int b = 0; b < M; ++b) acc[b] = 0.f;
    for (int64_t k0 = (int64_t)lane * 8; k0 < K; k0 += 64 * 8) {
      gus16x8 w8 = *reinterpret_cast<const gus16x8*>(wr + k0);
#pragma unroll
      for (int b = 0; b < M; ++b) {
        gus16x8 x8 = *reinterpret_cast<const gus16x8*>(x + (int64_t)b * K + k0);
#pragma unroll
        for (int p = 0; p < 4; ++p) {
          gus16x2 wp = {w8[2 * p], w8[2 * p + 1]};
          gus16x2 xp = {x8[2 * p], x8[2 * p + 1]};
          acc[b] = __builtin_amdgcn_fdot2_f32_bf16(wp, xp, acc[b], false);
        }
      }
    }
#pragma unroll
    for (int b = 0; b < M; ++b) {
#pragma unroll
      for (int off = 32; off > 0; off >>= 1) acc[b] += __shfl_down(acc[b], off, 64);
    }
    if (lane == 0) {
      float bv = 0.f;
      if (bias != nullptr) {
        union { ushort u; } ub{bias[row]};
        unsigned wide = ((unsigned)ub.u) << 16;
        bv = *reinterpret_cast<float*>(&wide);
      }
#pragma unroll
      for (int b = 0; b < M; ++b) y[(int64_t)b * N + row] = gvf2bf(acc[b] + bv);
    }
  }
}

extern "C" hipError_t launch_gemv_bf16(const void* w, const void* x, const void* bias, void* y,
                                       long long N, long long K, int M, hipStream_t stream) {
  const int n_blocks = (int)std::min<long long>((N + 3) / 4, 4096);
  dim3 g(n_blocks), b(256);
#define GEMV_CASE(MM) \
  hipLaunchKernelGGL((gemv_bf16_kernel<MM>), g, b, 0, stream, (const ushort*)w, \
                     (const ushort*)x, (const ushort*)bias, (ushort*)y, (int64_t)N, (int64_t)K)
  switch (M) {
    case 1: GEMV_CASE(1); break;
    case 2: GEMV_CASE(2); break;
    case 3: GEMV_CASE(3); break;
    case 4: GEMV_CASE(4); break;
    case 5: GEMV_CASE(5); break;
    case 6: GEMV_CASE(6); break;
    case 7: GEMV_CASE(7); break;
    case 8: GEMV_CASE(8); break;
    default: return hipErrorInvalidValue;
  }
#undef GEMV_CASE
  return hipGetLastError();
}
