// Fused flash-attention forward for gfx950 (CDNA4), bf16, head_dim 64/128.
//
// Replaces the blockwise torch-ops forward in ops/attention.py (same
// semantics: returns out AND the natural-log logsumexp, so the existing
// logsumexp-recompute backward keeps working unchanged).
//
// Structure (FA2 on CDNA4):
//   grid = (ceil(Sq/128), B*H); block = 256 threads = 4 waves.
//   Each wave owns 32 query rows, so every softmax row statistic is a
//   16-lane shuffle reduction — no cross-wave LDS reductions.
//   Per key-block of 128: S = Q·K^T on MFMA 16x16x32 (fragment mapping
//   verified by benchmarks/mfma_layout_probe.hip: k = (lane>>4)*8+j,
//   C row = (lane>>4)*4+j, col = lane&15), online softmax in fp32
//   registers (base-2 domain: v_exp_f32 IS exp2), P staged wave-private
//   through LDS to re-shape C-layout -> A-layout, then O += P·V on MFMA
//   with V held transposed in LDS so B-fragment reads are contiguous
//   ds_read_b128s. LDS rows padded +8 bf16 to break bank conflicts.
//
// Q stays in registers for the whole block (loaded once); K/V tiles are
// staged per key-block. Softmax statistics m (running max, base-2) and
// l (denominator) live per (row) in 4 registers per lane.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace {

typedef __bf16 abf16;
typedef abf16 bf16x8a __attribute__((ext_vector_type(8)));
typedef float f32x4a __attribute__((ext_vector_type(4)));

__device__ __forceinline__ ushort af2bf(float f) {
  union { float f; unsigned int i; } cv;
  cv.f = f;
  if ((cv.i & 0x7F800000u) == 0x7F800000u) return (ushort)(cv.i >> 16) | (ushort)((cv.i & 0xFFFFu) ? 0x40 : 0);
  cv.i += 0x7FFFu + ((cv.i >> 16) & 1u);
  return (ushort)(cv.i >> 16);
}

constexpr float kLog2e = 1.4426950408889634f;
constexpr float kLn2 = 0.6931471805599453f;

}  // namespace

// D = head_dim (64 or 128). q:[B,H,Sq,D] k,v:[B,H,Sk,D] bf16 contiguous.
// out:[B,H,Sq,D] bf16; lse:[B,H,Sq] fp32 natural-log.
// past: causal offset — query i attends keys <= past + i.
template <int D>
__global__ __launch_bounds__(256) void fa_fwd_kernel(
    const abf16* __restrict__ q, const abf16* __restrict__ k,
    const abf16* __restrict__ v, ushort* __restrict__ out,
    float* __restrict__ lse, int Sq, int Sk, int past, int causal,
    float scale) {
  constexpr int BM = 128, BN = 128;
  constexpr int KP = D + 8;    // padded K-tile row stride (bf16)
  constexpr int NP = BN + 8;   // padded VT / P row stride (bf16)
  extern __shared__ char smem[];
  abf16* Ks = reinterpret_cast<abf16*>(smem);                 // [BN][KP]
  abf16* VTs = Ks + BN * KP;                                  // [D][NP]
  abf16* Ps = VTs + D * NP;                                   // [BM][NP]

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int q0 = blockIdx.x * BM;
  const int64_t bh = blockIdx.y;  // b*H + h
  const abf16* qb = q + (bh * Sq) * (int64_t)D;
  const abf16* kb = k + (bh * Sk) * (int64_t)D;
  const abf16* vb = v + (bh * Sk) * (int64_t)D;

  // --- Q fragments, loaded once: wave rows [wave*32, wave*32+32) ---
  // A-frag (mi = 16-row tile, kc = 32-wide k chunk): lane holds row
  // base+(lane&15), k = kc*32 + (lane>>4)*8 + 0..7 -> one 16 B global read.
  bf16x8a qf[2][D / 32];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
    int qr = q0 + wave * 32 + mi * 16 + (lane & 15);
    if (qr >= Sq) qr = Sq - 1;  // clamp (padded rows never stored)
#pragma unroll
    for (int kc = 0; kc < D / 32; ++kc)
      qf[mi][kc] = *reinterpret_cast<const bf16x8a*>(qb + (int64_t)qr * D + kc * 32 + (lane >> 4) * 8);
  }

  f32x4a oacc[2][D / 16];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int nd = 0; nd < D / 16; ++nd) oacc[mi][nd] = f32x4a{};
  float m_run[2][4], l_run[2][4];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int j = 0; j < 4; ++j) { m_run[mi][j] = -INFINITY; l_run[mi][j] = 0.f; }

  const float c = scale * kLog2e;  // fold scale into the base-2 domain
  const int k_hi = causal ? min(Sk, past + q0 + BM) : Sk;

  for (int kb0 = 0; kb0 < k_hi; kb0 += BN) {
    // --- stage K [BN][D] row-major and V^T [D][BN] into LDS ---
    // 256 threads x 16 B chunks; global reads coalesce along rows.
    {
      constexpr int chunks = BN * D / 8;         // 16 B chunks in the tile
#pragma unroll
      for (int it = 0; it < chunks / 256; ++it) {
        const int cid = tid + it * 256;
        const int row = cid / (D / 8);           // key within tile
        const int col8 = (cid % (D / 8)) * 8;    // d offset
        int64_t key = kb0 + row;
        if (key >= Sk) key = Sk - 1;             // clamp (masked below)
        bf16x8a kv8 = *reinterpret_cast<const bf16x8a*>(kb + key * D + col8);
        *reinterpret_cast<bf16x8a*>(Ks + row * KP + col8) = kv8;
        bf16x8a vv8 = *reinterpret_cast<const bf16x8a*>(vb + key * D + col8);
#pragma unroll
        for (int j = 0; j < 8; ++j) VTs[(col8 + j) * NP + row] = vv8[j];
      }
    }
    __syncthreads();

    // --- S = Q K^T fragments: sacc[mi][ni] covers rows 16, cols 16 ---
    f32x4a sacc[2][8];
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < 8; ++ni) sacc[mi][ni] = f32x4a{};
#pragma unroll
    for (int ni = 0; ni < 8; ++ni) {
#pragma unroll
      for (int kc = 0; kc < D / 32; ++kc) {
        bf16x8a bf = *reinterpret_cast<const bf16x8a*>(
            Ks + (ni * 16 + (lane & 15)) * KP + kc * 32 + (lane >> 4) * 8);
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
          sacc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[mi][kc], bf, sacc[mi][ni], 0, 0, 0);
      }
    }

    // --- mask (causal upper edge and Sk tail) in the RAW score domain ---
    const bool edge = (kb0 + BN > Sk) || (causal && kb0 + BN > past + q0);
    if (edge) {
#pragma unroll
      for (int ni = 0; ni < 8; ++ni) {
        const int kc = kb0 + ni * 16 + (lane & 15);
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            const int qr = q0 + wave * 32 + mi * 16 + (lane >> 4) * 4 + j;
            if (kc >= Sk || (causal && kc > past + qr)) sacc[mi][ni][j] = -INFINITY;
          }
        }
      }
    }

    // --- online softmax (base-2): every row lives in 16 lanes ---
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float bm = -INFINITY;
#pragma unroll
        for (int ni = 0; ni < 8; ++ni) bm = fmaxf(bm, sacc[mi][ni][j]);
        bm *= c;  // c > 0 so max commutes with the scale fold
#pragma unroll
        for (int x = 1; x < 16; x <<= 1) bm = fmaxf(bm, __shfl_xor(bm, x, 64));
        const float new_m = fmaxf(m_run[mi][j], bm);
        // every processed row has >= 1 unmasked key (causal rows see key 0)
        const float corr = (m_run[mi][j] == -INFINITY) ? 0.f : exp2f(m_run[mi][j] - new_m);
        m_run[mi][j] = new_m;
        float rs = 0.f;
#pragma unroll
        for (int ni = 0; ni < 8; ++ni) {
          const float p = exp2f(sacc[mi][ni][j] * c - new_m);
          sacc[mi][ni][j] = p;
          rs += p;
        }
#pragma unroll
        for (int x = 1; x < 16; x <<= 1) rs += __shfl_xor(rs, x, 64);
        l_run[mi][j] = l_run[mi][j] * corr + rs;
#pragma unroll
        for (int nd = 0; nd < D / 16; ++nd) oacc[mi][nd][j] *= corr;
      }
    }

    // --- P -> LDS (wave-private rows; C-layout scatter, bf16) ---
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < 8; ++ni)
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int row = wave * 32 + mi * 16 + (lane >> 4) * 4 + j;
          ushort pb = af2bf(sacc[mi][ni][j]);
          Ps[row * NP + ni * 16 + (lane & 15)] = *reinterpret_cast<abf16*>(&pb);
        }
    // same-wave write->read: the compiler orders the ds ops (no barrier —
    // no wave reads another wave's P rows)

    // --- O += P V : A = P (rows, k=key), B = V^T rows are d ---
#pragma unroll
    for (int kc = 0; kc < BN / 32; ++kc) {
      bf16x8a pf[2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
        pf[mi] = *reinterpret_cast<const bf16x8a*>(
            Ps + (wave * 32 + mi * 16 + (lane & 15)) * NP + kc * 32 + (lane >> 4) * 8);
#pragma unroll
      for (int nd = 0; nd < D / 16; ++nd) {
        bf16x8a vf = *reinterpret_cast<const bf16x8a*>(
            VTs + (nd * 16 + (lane & 15)) * NP + kc * 32 + (lane >> 4) * 8);
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
          oacc[mi][nd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf[mi], vf, oacc[mi][nd], 0, 0, 0);
      }
    }
    __syncthreads();  // next iteration restages Ks/VTs
  }

  // --- epilogue: out = O/l (bf16), lse = (m + log2 l) * ln2 (natural) ---
  ushort* ob = out + (bh * Sq) * (int64_t)D;
  float* lb = lse + bh * Sq;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int qr = q0 + wave * 32 + mi * 16 + (lane >> 4) * 4 + j;
      if (qr >= Sq) continue;
      const float inv_l = 1.f / fmaxf(l_run[mi][j], 1e-30f);
#pragma unroll
      for (int nd = 0; nd < D / 16; ++nd)
        ob[(int64_t)qr * D + nd * 16 + (lane & 15)] = af2bf(oacc[mi][nd][j] * inv_l);
      if ((lane & 15) == 0)
        lb[qr] = (m_run[mi][j] + __log2f(fmaxf(l_run[mi][j], 1e-30f))) * kLn2;
    }
  }
}

template __global__ void fa_fwd_kernel<64>(const abf16*, const abf16*, const abf16*,
                                           ushort*, float*, int, int, int, int, float);
template __global__ void fa_fwd_kernel<128>(const abf16*, const abf16*, const abf16*,
                                            ushort*, float*, int, int, int, int, float);

namespace {

template <int D>
int fa_lds_bytes() {
  return (128 * (D + 8) + D * 136 + 128 * 136) * 2;
}

template <int D>
hipError_t launch_impl(const void* q, const void* k, const void* v, void* out, float* lse,
                       int64_t bh, int Sq, int Sk, int past, int causal, float scale,
                       hipStream_t stream) {
  static bool attr_set = false;
  const int lds = fa_lds_bytes<D>();
  if (!attr_set) {
    hipError_t e = hipFuncSetAttribute(reinterpret_cast<const void*>(&fa_fwd_kernel<D>),
                                       hipFuncAttributeMaxDynamicSharedMemorySize, lds);
    if (e != hipSuccess) return e;
    attr_set = true;
  }
  dim3 grid((Sq + 127) / 128, (unsigned)bh);
  hipLaunchKernelGGL(fa_fwd_kernel<D>, grid, dim3(256), lds, stream,
                     reinterpret_cast<const abf16*>(q), reinterpret_cast<const abf16*>(k),
                     reinterpret_cast<const abf16*>(v), reinterpret_cast<ushort*>(out),
                     lse, Sq, Sk, past, causal, scale);
  return hipGetLastError();
}

}  // namespace

// raw launcher used by bindings.hip (template instantiations stay local)
hipError_t launch_fa_fwd(const void* q, const void* k, const void* v, void* out, float* lse,
                         int64_t batch_heads, int Sq, int Sk, int head_dim, int past,
                         int causal, float scale, hipStream_t stream) {
  if (head_dim == 64)
    return launch_impl<64>(q, k, v, out, lse, batch_heads, Sq, Sk, past, causal, scale, stream);
  if (head_dim == 128)
    return launch_impl<128>(q, k, v, out, lse, batch_heads, Sq, Sk, past, causal, scale, stream);
  return hipErrorInvalidValue;
}
