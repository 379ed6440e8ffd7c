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
#include <cstdlib>
#include <cstring>
#include <hip/hip_bf16.h>

#include "attention.h"

namespace {

typedef __bf16 abf16;
typedef abf16 bf16x8a __attribute__((ext_vector_type(8)));
typedef float f32x4a __attribute__((ext_vector_type(4)));

// f32 -> bf16 through the HARDWARE packed convert: a plain static_cast
// compiles to v_cvt_pk_bf16_f32 (RNE, NaN-correct) and the compiler packs
// adjacent pairs — the previous branchy bit-twiddled round-to-nearest cost
// ~7 VALU ops per element and was the top VALU source in the disassembly
// (2983 VALU vs 128 MFMA in fa_fwd<128>, docs/NEXT_STEPS round 1).
__device__ __forceinline__ ushort af2bf(float f) {
  abf16 h = (abf16)f;
  return *reinterpret_cast<ushort*>(&h);
}


// bit view of one lane of a bf16 vector (ext_vector elements have no address)
__device__ __forceinline__ unsigned bfbits(abf16 h) {
  union { abf16 h; ushort u; } cv;
  cv.h = h;
  return (unsigned)cv.u;
}

constexpr float kLog2e = 1.4426950408889634f;
constexpr float kLn2 = 0.6931471805599453f;

}  // namespace

namespace {
// exp2f lowers to ldexp+v_exp range-fixup pairs without -ffast-math; the
// softmax arguments are bounded (s - m <= defer-threshold, s - lse <= 0 on
// real rows), so the raw hardware v_exp_f32 is exact where it matters.
__device__ __forceinline__ float fast_exp2(float x) { return __builtin_amdgcn_exp2f(x); }
}  // namespace

// D = head_dim (64 or 128). q:[B,Hq,Sq,D] k,v:[B,Hkv,Sk,D] bf16 views
// (strides in elements, dim 3 contiguous). out: written through sO (BSHD
// storage); lse:[B,Hq,Sq] fp32 natural-log contiguous.
// past: causal offset — query i attends keys <= past + i.
template <int D>
__global__ __launch_bounds__(256, 2) void fa_fwd_kernel(
    const abf16* __restrict__ q, const abf16* __restrict__ k,
    const abf16* __restrict__ v, ushort* __restrict__ out,
    float* __restrict__ lse, int Sq, int Sk, int past, int causal,
    float scale, int Hq, int Hkv, Str3 sQ, Str3 sK, Str3 sV, Str3 sO) {
  // BN=128 measured faster than BN=64 (10413 vs 10239 tok/s on Llama-8B
  // b4 no-AC): the extra occupancy at 54 KB LDS is VGPR-capped anyway and
  // BN=64 doubles the barrier/softmax-reduction overhead per key.
  constexpr int BM = 128, BN = 128;
  constexpr int KP = D + 8;    // padded K-tile row stride (bf16)
  constexpr int NP = BN + 8;   // padded VT / P row stride (bf16)
  extern __shared__ char smem[];
  abf16* Ks = reinterpret_cast<abf16*>(smem);                 // [BN][KP]
  abf16* VTs = Ks + BN * KP;                                  // [D][NP]
  // P staging ALIASES the K tile at D=128 (exactly BM*NP == BN*KP bf16):
  // K is fully consumed by the QK^T MFMAs, so after a barrier the same
  // LDS holds P. This cuts the block from 102 KB to 68 KB — 2 blocks/CU
  // instead of 1, which is the occupancy this kernel was starved of
  // (round-1 PMC: ~80% stall at occupancy 1, profiles/llama8b_fa_pmc.md).
  abf16* Ps = (BM * NP <= BN * KP) ? Ks : VTs + D * NP;       // [BM][NP]

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int q0 = blockIdx.x * BM;
  const int64_t bh = blockIdx.y;  // b*Hq + h
  const int64_t b = bh / Hq, h = bh % Hq, hk = h / (Hq / Hkv);
  const abf16* qb = q + b * sQ.b + h * sQ.h;
  const abf16* kb = k + b * sK.b + hk * sK.h;
  const abf16* vb = v + b * sV.b + hk * sV.h;

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
      qf[mi][kc] = *reinterpret_cast<const bf16x8a*>(qb + (int64_t)qr * sQ.s + kc * 32 + (lane >> 4) * 8);
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
    // Each thread owns a PAIR of adjacent key rows: the V transpose then
    // writes packed 2-key b32 columns (8 ds_write_b32 per pair) instead of
    // 16 scalar ds_write_b16 — half the LDS-write ops of the scatter.
    {
      constexpr int pair_chunks = (BN / 2) * (D / 8);
#pragma unroll
      for (int it = 0; it < pair_chunks / 256; ++it) {
        const int cid = tid + it * 256;
        const int r0 = (cid / (D / 8)) * 2;      // even key within tile
        const int col8 = (cid % (D / 8)) * 8;    // d offset
        int64_t key0 = kb0 + r0, key1 = kb0 + r0 + 1;
        if (key0 >= Sk) key0 = Sk - 1;           // clamp (masked below)
        if (key1 >= Sk) key1 = Sk - 1;
        *reinterpret_cast<bf16x8a*>(Ks + r0 * KP + col8) =
            *reinterpret_cast<const bf16x8a*>(kb + key0 * sK.s + col8);
        *reinterpret_cast<bf16x8a*>(Ks + (r0 + 1) * KP + col8) =
            *reinterpret_cast<const bf16x8a*>(kb + key1 * sK.s + col8);
        bf16x8a v0 = *reinterpret_cast<const bf16x8a*>(vb + key0 * sV.s + col8);
        bf16x8a v1 = *reinterpret_cast<const bf16x8a*>(vb + key1 * sV.s + col8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const unsigned pk = bfbits(v0[j]) | (bfbits(v1[j]) << 16);
          *reinterpret_cast<unsigned*>(VTs + (col8 + j) * NP + r0) = pk;
        }
      }
    }
    __syncthreads();

    // --- S = Q K^T fragments: sacc[mi][ni] covers rows 16, cols 16 ---
    f32x4a sacc[2][BN / 16];
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < BN / 16; ++ni) sacc[mi][ni] = f32x4a{};
    __builtin_amdgcn_s_setprio(1);  // favor this wave while the MFMA cluster runs
#pragma unroll
    for (int ni = 0; ni < BN / 16; ++ni) {
#pragma unroll
      for (int kc = 0; kc < D / 32; ++kc) {
        bf16x8a bf = *reinterpret_cast<const bf16x8a*>(
            Ks + (ni * 16 + (lane & 15)) * KP + kc * 32 + (lane >> 4) * 8);
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
          sacc[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[mi][kc], bf, sacc[mi][ni], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);

    // --- mask (causal upper edge and Sk tail) in the RAW score domain ---
    // row/col indices are affine in the loop counters — hoisted bases, one
    // add per element instead of a 4-term recomputation
    const bool edge = (kb0 + BN > Sk) || (causal && kb0 + BN > past + q0);
    if (edge) {
      const int kc0 = kb0 + (lane & 15);
      const int qr0 = q0 + wave * 32 + (lane >> 4) * 4;
#pragma unroll
      for (int ni = 0; ni < BN / 16; ++ni) {
        const int kc = kc0 + ni * 16;
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          const int qb16 = qr0 + mi * 16;
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            if (kc >= Sk || (causal && kc > past + qb16 + j)) sacc[mi][ni][j] = -INFINITY;
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
        for (int ni = 0; ni < BN / 16; ++ni) bm = fmaxf(bm, sacc[mi][ni][j]);
        bm *= c;  // c > 0 so max commutes with the scale fold
#pragma unroll
        for (int x = 1; x < 16; x <<= 1) bm = fmaxf(bm, __shfl_xor(bm, x, 64));
        const float new_m = fmaxf(m_run[mi][j], bm);
        // every processed row has >= 1 unmasked key (causal rows see key 0)
        const float corr = (m_run[mi][j] == -INFINITY) ? 0.f : fast_exp2(m_run[mi][j] - new_m);
        m_run[mi][j] = new_m;
        float rs = 0.f;
#pragma unroll
        for (int ni = 0; ni < BN / 16; ++ni) {
          const float p = fast_exp2(sacc[mi][ni][j] * c - new_m);
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
    // Ps aliases Ks at D=128: EVERY wave must be done with its QK^T reads
    // of the K tile before any wave's P lands on top of it.
    __syncthreads();
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        abf16* prow = Ps + (wave * 32 + mi * 16 + (lane >> 4) * 4 + j) * NP + (lane & 15);
#pragma unroll
        for (int ni = 0; ni < BN / 16; ++ni) prow[ni * 16] = (abf16)sacc[mi][ni][j];
      }
    // same-wave write->read: the compiler orders the ds ops (no barrier —
    // no wave reads another wave's P rows)

    // --- O += P V : A = P (rows, k=key), B = V^T rows are d ---
    __builtin_amdgcn_s_setprio(1);
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
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();  // next iteration restages Ks/VTs
  }

  // --- epilogue: out = O/l (bf16), lse = (m + log2 l) * ln2 (natural) ---
  ushort* ob = out + b * sO.b + h * sO.h;
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
        ob[(int64_t)qr * sO.s + nd * 16 + (lane & 15)] = af2bf(oacc[mi][nd][j] * inv_l);
      if ((lane & 15) == 0)
        lb[qr] = (m_run[mi][j] + __log2f(fmaxf(l_run[mi][j], 1e-30f))) * kLn2;
    }
  }
}

// ---------------------------------------------------------------------------
// Swapped-QK^T forward (CDNA4 8-warp 32x32 ladder, D=128).
//
// Each of 8 waves owns 32 query rows (BM=256), key tiles of BN=64.
// S^T is computed as mfma(K, Q) on 32x32x16 MFMA so each lane holds a
// full half-row of P for ONE query row (q = lane&31): row statistics are
// 31 in-register fmax/adds + ONE permlane32_swap — no cross-lane shuffles
// ladders, and P never round-trips through LDS: the bf16 pair-pack +
// permlane32_swap relayout (verified by benchmarks/fa_swapped_probe.hip)
// turns the C-layout scores directly into PV's A-fragments.
// Softmax runs in the base-2 domain; defer-max (skip the O rescale while
// the running max grows by < 8) removes most O-wide rescale passes.
// ---------------------------------------------------------------------------

namespace {
typedef abf16 bf16x2a __attribute__((ext_vector_type(2)));
typedef float f32x16a __attribute__((ext_vector_type(16)));
typedef unsigned uint2a __attribute__((ext_vector_type(2)));

__device__ __forceinline__ unsigned pack_bf16_pair(float lo, float hi) {
  union { bf16x2a h; unsigned u; } cv;
  cv.h = bf16x2a{(abf16)lo, (abf16)hi};
  return cv.u;
}
}  // namespace

typedef abf16 bf16x4t __attribute__((ext_vector_type(4)));

// TR=true stores the V tile as [k/4][d/16][4][16] subtiles and reads PV's
// B-fragments with ds_read_b64_tr_b16 (hardware transpose; each 16-lane
// group's addresses tile one 4x16 subtile window — fa_swapped_probe
// stage E), replacing the pair-packed V^T staging scatter.
template <int D, bool TR>
__global__ __launch_bounds__(512, 1) void fa_fwd_swapped_kernel(
    const abf16* __restrict__ q, const abf16* __restrict__ k,
    const abf16* __restrict__ v, ushort* __restrict__ out,
    float* __restrict__ lse, int Sq, int Sk, int past, int causal,
    float scale, int Hq, int Hkv, Str3 sQ, Str3 sK, Str3 sV, Str3 sO) {
  static_assert(D == 64 || D == 128, "swapped ladder supports head_dim 64/128");
  constexpr int BM = 256, BN = 64;
  // +8 padding measured BETTER than the XOR-chunk swizzle here (630 vs
  // 591 TF/s non-causal): the b128 fragment reads are 4-way-conflict
  // bounded either way, and padding keeps the addressing immediate-only.
  constexpr int KP = D + 8;   // K tile row stride
  constexpr int NP = BN + 8;  // V^T row stride
  // DOUBLE-BUFFERED tiles: one barrier per iteration instead of two, so
  // the two co-resident waves per SIMD can drift a phase apart and
  // interleave softmax VALU with the other wave's MFMA clusters.
  constexpr int TILE = BN * KP + D * NP;  // one K+V buffer (elements)
  extern __shared__ char smem[];
  abf16* tiles = reinterpret_cast<abf16*>(smem);  // [2][TILE]

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;     // this lane's query row within the wave tile
  // causal blocks are launched heavy-first (largest q0 has the most key
  // tiles): the light diagonal-top blocks then pack the scheduling tail
  const int q0 = (causal ? (int)(gridDim.x - 1 - blockIdx.x) : (int)blockIdx.x) * BM;
  const int64_t bh = blockIdx.y;
  const int64_t b = bh / Hq, h = bh % Hq, hk = h / (Hq / Hkv);
  const abf16* qb = q + b * sQ.b + h * sQ.h;
  const abf16* kb = k + b * sK.b + hk * sK.h;
  const abf16* vb = v + b * sV.b + hk * sV.h;

  // Q as B-operand fragments, loaded once: lane holds Q row
  // q0 + wave*32 + col, chunk t covers d = t*16 + hi*8 + 0..7
  int qr_mine = q0 + wave * 32 + col;
  if (qr_mine >= Sq) qr_mine = Sq - 1;  // clamp; padded rows never stored
  // fold c = scale*log2e INTO the Q fragments once (64 converts here)
  // instead of scaling every score every tile (32 VALU/lane/tile): S then
  // lands directly in the base-2 domain. bf16 re-round only shifts the
  // exponent (c is a pure magnitude), so precision is unchanged.
  const float c = scale * kLog2e;
  bf16x8a qf[D / 16];
#pragma unroll
  for (int t = 0; t < D / 16; ++t) {
    bf16x8a raw = *reinterpret_cast<const bf16x8a*>(qb + (int64_t)qr_mine * sQ.s + t * 16 + hi * 8);
#pragma unroll
    for (int j = 0; j < 8; ++j) raw[j] = (abf16)((float)raw[j] * c);
    qf[t] = raw;
  }

  f32x16a oacc[D / 32] = {};
  float m_run = -INFINITY;  // base-2 scaled domain, row = col
  float l_run = 0.f;        // THIS lane's half-row partial denominator
  const int k_hi = causal ? min(Sk, past + q0 + BM) : Sk;
  // this wave's own causal horizon: rows q0+wave*32 .. +32
  const int k_hi_wave = causal ? min(Sk, past + q0 + wave * 32 + 32) : Sk;

  // async-STAGE (T14): tile kb0+BN's global loads are ISSUED right after
  // this tile's QK^T MFMAs and land in registers while softmax/PV run;
  // the LDS write happens after the end-of-iteration barrier.
  const int kr_row = tid / (D / 8);
  const int kr_col8 = (tid % (D / 8)) * 8;
  const int vr_r0 = (tid / (D / 8)) * 2;
  // K tile = BN*(D/8) b128 chunks over 512 threads: 2 per thread at D=128,
  // 1 at D=64 (a second pass would walk past the 64-key tile)
  constexpr int KIT = BN * (D / 8) / 512;
  bf16x8a krg[KIT], vrg0, vrg1;
  auto issue_loads = [&](int kt0) {
#pragma unroll
    for (int it = 0; it < KIT; ++it) {
      int64_t key = kt0 + kr_row + it * (512 / (D / 8));
      if (key >= Sk) key = Sk - 1;  // clamp (masked below)
      krg[it] = *reinterpret_cast<const bf16x8a*>(kb + key * sK.s + kr_col8);
    }
    if (vr_r0 < 64) {  // D=64: only BN/2*(D/8)=256 pair-chunks
      int64_t key0 = kt0 + vr_r0, key1 = kt0 + vr_r0 + 1;
      if (key0 >= Sk) key0 = Sk - 1;
      if (key1 >= Sk) key1 = Sk - 1;
      vrg0 = *reinterpret_cast<const bf16x8a*>(vb + key0 * sV.s + kr_col8);
      vrg1 = *reinterpret_cast<const bf16x8a*>(vb + key1 * sV.s + kr_col8);
    }
  };
  auto write_tile = [&](int buf) {
    abf16* Ks = tiles + buf * TILE;
    abf16* VTs = Ks + BN * KP;
#pragma unroll
    for (int it = 0; it < KIT; ++it)
      *reinterpret_cast<bf16x8a*>(Ks + (kr_row + it * (512 / (D / 8))) * KP + kr_col8) = krg[it];
    if (TR) {
      // subtiled [k/4][d/16][4][16]: natural vectorized stores, no packing
      if (vr_r0 < 64) {
        const int db = (kr_col8 >> 4) * 64 + (kr_col8 & 15);
        *reinterpret_cast<bf16x8a*>(VTs + ((vr_r0 >> 2) * (D / 16)) * 64 + db +
                                    (vr_r0 & 3) * 16) = vrg0;
        *reinterpret_cast<bf16x8a*>(VTs + (((vr_r0 + 1) >> 2) * (D / 16)) * 64 + db +
                                    ((vr_r0 + 1) & 3) * 16) = vrg1;
      }
    } else {
      if (vr_r0 < 64) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const unsigned pk = bfbits(vrg0[j]) | (bfbits(vrg1[j]) << 16);
          *reinterpret_cast<unsigned*>(VTs + (kr_col8 + j) * NP + vr_r0) = pk;
        }
      }
    }
  };

  issue_loads(0);
  write_tile(0);
  int buf = 0;
  for (int kb0 = 0; kb0 < k_hi; kb0 += BN) {
    __syncthreads();  // staging of `buf` done; prior reads of buf^1 done
    const abf16* Ks = tiles + buf * TILE;
    const abf16* VTs = Ks + BN * KP;
    if (kb0 + BN < k_hi) issue_loads(kb0 + BN);

    f32x16a sacc[2] = {f32x16a{}, f32x16a{}};
    const bool live = kb0 < k_hi_wave;  // fully-masked tiles: no compute
    if (live) {
      // --- S^T = mfma(K, Q): sacc[kt][r] = S[col][kt*32 + crow(r,hi)] ---
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kt = 0; kt < 2; ++kt)
#pragma unroll
        for (int t = 0; t < D / 16; ++t) {
          bf16x8a kf = *reinterpret_cast<const bf16x8a*>(
              Ks + (kt * 32 + col) * KP + t * 16 + hi * 8);
          sacc[kt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[t], sacc[kt], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);
    }
    if (kb0 + BN < k_hi) write_tile(buf ^ 1);  // regs -> other buffer
    if (live) {
      // --- mask (causal diagonal / Sk tail) in the raw domain ---
      const int qabs = q0 + wave * 32 + col;
      const bool edge = (kb0 + BN > Sk) || (causal && kb0 + BN > past + q0 + wave * 32);
      if (edge) {
#pragma unroll
        for (int kt = 0; kt < 2; ++kt) {
          const int kbase = kb0 + kt * 32 + 4 * hi;
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            const int kabs = kbase + (r & 3) + 8 * (r >> 2);
            if (kabs >= Sk || (causal && kabs > past + qabs)) sacc[kt][r] = -INFINITY;
          }
        }
      }

      // --- in-register softmax: 31 fmax + one permlane32_swap ---
      float pmax = sacc[0][0];
#pragma unroll
      for (int r = 1; r < 16; ++r) pmax = fmaxf(pmax, sacc[0][r]);
#pragma unroll
      for (int r = 0; r < 16; ++r) pmax = fmaxf(pmax, sacc[1][r]);
      {
        uint2a sw = __builtin_amdgcn_permlane32_swap(__float_as_uint(pmax),
                                                     __float_as_uint(pmax), false, false);
        pmax = fmaxf(__uint_as_float(sw.x), __uint_as_float(sw.y));
      }
      // defer-max: keep m_run while the max grows slowly (P bounded by 2^8)
      const bool rescale = !__all(pmax - m_run <= 8.f);
      if (rescale) {
        const float new_m = fmaxf(m_run, pmax);
        const float corr = (m_run == -INFINITY) ? 0.f : fast_exp2(m_run - new_m);
        m_run = new_m;
        l_run *= corr;
        // redistribute corr to O rows (row of value r is crow(r,hi))
        float cr[16];
#pragma unroll
        for (int r = 0; r < 16; ++r)
          cr[r] = __shfl(corr, (r & 3) + 8 * (r >> 2) + 4 * hi, 64);
#pragma unroll
        for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
          for (int r = 0; r < 16; ++r) oacc[dt][r] *= cr[r];
      }

      // --- P = exp2(S*c - m), accumulate this lane's half-row sum ---
      float rs = 0.f;
#pragma unroll
      for (int kt = 0; kt < 2; ++kt)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float pv = fast_exp2(sacc[kt][r] - m_run);
          sacc[kt][r] = pv;
          rs += pv;
        }
      l_run += rs;

      // --- relayout P into PV A-fragments (pack + permlane32_swap) ---
      unsigned pa[4][4] __attribute__((aligned(16)));
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        const f32x16a& pp = sacc[ks >> 1];
        const int base = 8 * (ks & 1);
        uint2a r01 = __builtin_amdgcn_permlane32_swap(
            pack_bf16_pair(pp[base + 0], pp[base + 1]),
            pack_bf16_pair(pp[base + 4], pp[base + 5]), false, false);
        uint2a r23 = __builtin_amdgcn_permlane32_swap(
            pack_bf16_pair(pp[base + 2], pp[base + 3]),
            pack_bf16_pair(pp[base + 6], pp[base + 7]), false, false);
        pa[ks][0] = r01.x;
        pa[ks][1] = r23.x;
        pa[ks][2] = r01.y;
        pa[ks][3] = r23.y;
      }

      // --- O += P V : B-frags are V^T rows (plain b128 reads) ---
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
        for (int ks = 0; ks < 4; ++ks) {
          bf16x8a vf;
          if (TR) {
            const unsigned a0 =
                64u * ((4 * ks + 2 * hi) * (D / 16) + 2 * dt + ((lane >> 4) & 1)) +
                (lane & 15) * 4;
            bf16x4t lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                (__attribute__((address_space(3))) bf16x4t*)(VTs + a0));
            bf16x4t hh = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                (__attribute__((address_space(3))) bf16x4t*)(VTs + a0 + 64 * (D / 16)));
            vf = bf16x8a{lo[0], lo[1], lo[2], lo[3], hh[0], hh[1], hh[2], hh[3]};
          } else {
            vf = *reinterpret_cast<const bf16x8a*>(
                VTs + (dt * 32 + col) * NP + ks * 16 + hi * 8);
          }
          oacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<bf16x8a*>(pa[ks]), vf, oacc[dt], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);
    }
    buf ^= 1;
  }

  // --- epilogue: combine half-row denominators, normalize, store ---
  {
    uint2a sw = __builtin_amdgcn_permlane32_swap(__float_as_uint(l_run),
                                                 __float_as_uint(l_run), false, false);
    l_run = __uint_as_float(sw.x) + __uint_as_float(sw.y);
  }
  const float inv_l_mine = 1.f / fmaxf(l_run, 1e-30f);
  float il[16];
#pragma unroll
  for (int r = 0; r < 16; ++r)
    il[r] = __shfl(inv_l_mine, (r & 3) + 8 * (r >> 2) + 4 * hi, 64);

  ushort* ob = out + b * sO.b + h * sO.h;
  float* lb = lse + bh * Sq;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qr = q0 + wave * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
    if (qr >= Sq) continue;
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt)
      ob[(int64_t)qr * sO.s + dt * 32 + col] = af2bf(oacc[dt][r] * il[r]);
  }
  const int qr_lse = q0 + wave * 32 + col;
  if (hi == 0 && qr_lse < Sq)
    lb[qr_lse] = (m_run + __log2f(fmaxf(l_run, 1e-30f))) * kLn2;
}

template __global__ void fa_fwd_swapped_kernel<128, false>(
    const abf16*, const abf16*, const abf16*, ushort*, float*, int, int, int, int, float,
    int, int, Str3, Str3, Str3, Str3);
template __global__ void fa_fwd_swapped_kernel<128, true>(
    const abf16*, const abf16*, const abf16*, ushort*, float*, int, int, int, int, float,
    int, int, Str3, Str3, Str3, Str3);
template __global__ void fa_fwd_swapped_kernel<64, false>(
    const abf16*, const abf16*, const abf16*, ushort*, float*, int, int, int, int, float,
    int, int, Str3, Str3, Str3, Str3);
template __global__ void fa_fwd_swapped_kernel<64, true>(
    const abf16*, const abf16*, const abf16*, ushort*, float*, int, int, int, int, float,
    int, int, Str3, Str3, Str3, Str3);

template __global__ void fa_fwd_kernel<64>(const abf16*, const abf16*, const abf16*,
                                           ushort*, float*, int, int, int, int, float,
                                           int, int, Str3, Str3, Str3, Str3);
template __global__ void fa_fwd_kernel<128>(const abf16*, const abf16*, const abf16*,
                                            ushort*, float*, int, int, int, int, float,
                                            int, int, Str3, Str3, Str3, Str3);

// ===========================================================================
// Backward: logsumexp-recompute, two passes (no atomics):
//   pass A (key-outer):  dk, dv — each block owns 128 keys, loops q-blocks
//   pass B (query-outer): dq    — each block owns 128 queries, loops k-blocks
// plus a Drow = rowsum(dout . out) precompute. All accumulation fp32 in
// registers across the whole inner loop; one bf16 store per grad element.
// Math (ops/attention.py backward):
//   P  = exp(S*scale − L);  dv = Pᵀ dO;  dP = dO Vᵀ
//   dS = P ⊙ (dP − Drow)·scale;  dq = dS K;  dk = dSᵀ Q
// ===========================================================================

__global__ __launch_bounds__(256) void fa_drow_kernel(
    const ushort* __restrict__ dout, const ushort* __restrict__ out,
    float* __restrict__ drow, int64_t n_rows, int D, int Hq, int Sq,
    Str3 sDo, Str3 sO) {
  const int wave = threadIdx.x >> 6, lane = threadIdx.x & 63;
  for (int64_t r = (int64_t)blockIdx.x * 4 + wave; r < n_rows; r += (int64_t)gridDim.x * 4) {
    const int64_t bb = r / ((int64_t)Hq * Sq), hh = (r / Sq) % Hq, ss = r % Sq;
    const ushort* dor = dout + bb * sDo.b + hh * sDo.h + ss * sDo.s;
    const ushort* orr = out + bb * sO.b + hh * sO.h + ss * sO.s;
    float s = 0.f;
    for (int d = lane; d < D; d += 64) {
      ushort a = dor[d], b = orr[d];
      s += (float)*reinterpret_cast<__hip_bfloat16*>(&a) * (float)*reinterpret_cast<__hip_bfloat16*>(&b);
    }
#pragma unroll
    for (int x = 1; x < 64; x <<= 1) s += __shfl_xor(s, x, 64);
    if (lane == 0) drow[r] = s;
  }
}

// pass A: dk/dv. Block owns keys [kb0, kb0+128) of one (b,h); wave w owns
// keys [w*32, w*32+32). Inner loop: 64-query tiles. K,V rows live in
// registers as MFMA A-fragments for the whole block; Q and dO tiles are
// staged in LDS in BOTH layouts (row-major for B-fragments of S^T/dP^T,
// transposed for B-fragments of dk/dv).
template <int D, int BQ>
__global__ __launch_bounds__(256, 2) void fa_bwd_dkdv_kernel(
    const abf16* __restrict__ q, const abf16* __restrict__ k,
    const abf16* __restrict__ v, const abf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ drow,
    ushort* __restrict__ dk, ushort* __restrict__ dv,
    int Sq, int Sk, int past, int causal, float scale,
    int Hq, int Hkv, Str3 sQ, Str3 sK, Str3 sV, Str3 sDo) {
  constexpr int BK = 128;
  constexpr int KP = D + 8, QP = BQ + 8;
  extern __shared__ char smem[];
  abf16* Qs = reinterpret_cast<abf16*>(smem);   // [BQ][KP]
  abf16* QTs = Qs + BQ * KP;                    // [D][QP]
  abf16* dOs = QTs + D * QP;                    // [BQ][KP]
  abf16* dOTs = dOs + BQ * KP;                  // [D][QP]
  abf16* Ws = dOTs + D * QP;                    // [BK][QP] P^T then dS^T
  float* Ls = reinterpret_cast<float*>(Ws + BK * QP);  // [BQ] lse*log2e
  float* Ds = Ls + BQ;                                 // [BQ] drow

  const int tid = threadIdx.x, wave = tid >> 6, lane = tid & 63;
  const int k0 = blockIdx.x * BK;
  const int64_t bh = blockIdx.y;
  const int64_t b = bh / Hq, h = bh % Hq, hk = h / (Hq / Hkv);
  const abf16* qb = q + b * sQ.b + h * sQ.h;
  const abf16* kb = k + b * sK.b + hk * sK.h;
  const abf16* vb = v + b * sV.b + hk * sV.h;
  const abf16* dob = dout + b * sDo.b + h * sDo.h;
  const float* lb = lse + bh * Sq;
  const float* db = drow + bh * Sq;

  // K,V rows of this wave as A-fragments (row = lane&15 within 16-row tile)
  bf16x8a kf[2][D / 32], vf[2][D / 32];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
    int kr = k0 + wave * 32 + mi * 16 + (lane & 15);
    if (kr >= Sk) kr = Sk - 1;
#pragma unroll
    for (int kc = 0; kc < D / 32; ++kc) {
      kf[mi][kc] = *reinterpret_cast<const bf16x8a*>(kb + (int64_t)kr * sK.s + kc * 32 + (lane >> 4) * 8);
      vf[mi][kc] = *reinterpret_cast<const bf16x8a*>(vb + (int64_t)kr * sV.s + kc * 32 + (lane >> 4) * 8);
    }
  }
  f32x4a dkacc[2][D / 16], dvacc[2][D / 16];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int nd = 0; nd < D / 16; ++nd) { dkacc[mi][nd] = f32x4a{}; dvacc[mi][nd] = f32x4a{}; }

  const float c = scale * kLog2e;
  const int q_lo = causal ? max(0, ((k0 - past) / BQ) * BQ) : 0;

  for (int q0 = q_lo; q0 < Sq; q0 += BQ) {
    // stage Q/dO tiles (both layouts) + L*log2e + Drow; transposed copies
    // write packed 2-row b32 columns (see the fwd staging comment)
    {
      constexpr int pair_chunks = (BQ / 2) * (D / 8);
#pragma unroll
      for (int it = 0; it < pair_chunks / 256 + (pair_chunks % 256 != 0); ++it) {
        const int cid = tid + it * 256;
        if (pair_chunks % 256 != 0 && cid >= pair_chunks) break;
        const int r0 = (cid / (D / 8)) * 2;
        const int col8 = (cid % (D / 8)) * 8;
        int64_t qr0 = q0 + r0, qr1 = q0 + r0 + 1;
        if (qr0 >= Sq) qr0 = Sq - 1;  // clamped; masked via p=0 below
        if (qr1 >= Sq) qr1 = Sq - 1;
        bf16x8a qa = *reinterpret_cast<const bf16x8a*>(qb + qr0 * sQ.s + col8);
        bf16x8a qc = *reinterpret_cast<const bf16x8a*>(qb + qr1 * sQ.s + col8);
        *reinterpret_cast<bf16x8a*>(Qs + r0 * KP + col8) = qa;
        *reinterpret_cast<bf16x8a*>(Qs + (r0 + 1) * KP + col8) = qc;
        bf16x8a da = *reinterpret_cast<const bf16x8a*>(dob + qr0 * sDo.s + col8);
        bf16x8a dc = *reinterpret_cast<const bf16x8a*>(dob + qr1 * sDo.s + col8);
        *reinterpret_cast<bf16x8a*>(dOs + r0 * KP + col8) = da;
        *reinterpret_cast<bf16x8a*>(dOs + (r0 + 1) * KP + col8) = dc;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          unsigned pq = bfbits(qa[j]) | (bfbits(qc[j]) << 16);
          *reinterpret_cast<unsigned*>(QTs + (col8 + j) * QP + r0) = pq;
          unsigned pd = bfbits(da[j]) | (bfbits(dc[j]) << 16);
          *reinterpret_cast<unsigned*>(dOTs + (col8 + j) * QP + r0) = pd;
        }
      }
      if (tid < BQ) {
        int64_t qr = q0 + tid;
        if (qr >= Sq) qr = Sq - 1;
        Ls[tid] = lb[qr] * kLog2e;
        Ds[tid] = db[qr];
      }
    }
    __syncthreads();

    // S^T = K Q^T ; dP^T = V dO^T  (both A@B^T on MFMA, contraction D)
    f32x4a st[2][BQ / 16], dpt[2][BQ / 16];
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < BQ / 16; ++ni) { st[mi][ni] = f32x4a{}; dpt[mi][ni] = f32x4a{}; }
#pragma unroll
    for (int ni = 0; ni < BQ / 16; ++ni) {
#pragma unroll
      for (int kc = 0; kc < D / 32; ++kc) {
        bf16x8a bq = *reinterpret_cast<const bf16x8a*>(
            Qs + (ni * 16 + (lane & 15)) * KP + kc * 32 + (lane >> 4) * 8);
        bf16x8a bd = *reinterpret_cast<const bf16x8a*>(
            dOs + (ni * 16 + (lane & 15)) * KP + kc * 32 + (lane >> 4) * 8);
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          st[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(kf[mi][kc], bq, st[mi][ni], 0, 0, 0);
          dpt[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(vf[mi][kc], bd, dpt[mi][ni], 0, 0, 0);
        }
      }
    }

    // P^T = exp2(S^T*c − L[q]) with causal/tail mask; stage P^T (bf16)
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int ni = 0; ni < BQ / 16; ++ni) {
        const int qcol = q0 + ni * 16 + (lane & 15);
        const float l2 = Ls[ni * 16 + (lane & 15)];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int kr = k0 + wave * 32 + mi * 16 + (lane >> 4) * 4 + j;
          const bool dead = qcol >= Sq || kr >= Sk || (causal && kr > past + qcol);
          const float p = dead ? 0.f : fast_exp2(st[mi][ni][j] * c - l2);
          st[mi][ni][j] = p;
          ushort pb = af2bf(p);
          Ws[(wave * 32 + mi * 16 + (lane >> 4) * 4 + j) * QP + ni * 16 + (lane & 15)] =
              *reinterpret_cast<abf16*>(&pb);
        }
      }
    }
    // dv += P^T dO (A-frags from Ws, B-frags from dO^T; contraction = q)
#pragma unroll
    for (int kc = 0; kc < BQ / 32; ++kc) {
      bf16x8a pf[2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
        pf[mi] = *reinterpret_cast<const bf16x8a*>(
            Ws + (wave * 32 + mi * 16 + (lane & 15)) * QP + kc * 32 + (lane >> 4) * 8);
#pragma unroll
      for (int nd = 0; nd < D / 16; ++nd) {
        bf16x8a bf = *reinterpret_cast<const bf16x8a*>(
            dOTs + (nd * 16 + (lane & 15)) * QP + kc * 32 + (lane >> 4) * 8);
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
          dvacc[mi][nd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(pf[mi], bf, dvacc[mi][nd], 0, 0, 0);
      }
    }

    // dS^T = P^T ⊙ (dP^T − Drow[q])·scale ; overwrite Ws (wave-private rows)
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int ni = 0; ni < BQ / 16; ++ni) {
        const float dr = Ds[ni * 16 + (lane & 15)];
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const float ds = st[mi][ni][j] * (dpt[mi][ni][j] - dr) * scale;
          ushort b = af2bf(ds);
          Ws[(wave * 32 + mi * 16 + (lane >> 4) * 4 + j) * QP + ni * 16 + (lane & 15)] =
              *reinterpret_cast<abf16*>(&b);
        }
      }
    }
    // dk += dS^T Q (B-frags from Q^T)
#pragma unroll
    for (int kc = 0; kc < BQ / 32; ++kc) {
      bf16x8a sf[2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
        sf[mi] = *reinterpret_cast<const bf16x8a*>(
            Ws + (wave * 32 + mi * 16 + (lane & 15)) * QP + kc * 32 + (lane >> 4) * 8);
#pragma unroll
      for (int nd = 0; nd < D / 16; ++nd) {
        bf16x8a bf = *reinterpret_cast<const bf16x8a*>(
            QTs + (nd * 16 + (lane & 15)) * QP + kc * 32 + (lane >> 4) * 8);
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
          dkacc[mi][nd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(sf[mi], bf, dkacc[mi][nd], 0, 0, 0);
      }
    }
    __syncthreads();  // next q-tile restages Qs/dOs
  }

  // partial dk/dv per Q-HEAD, contiguous [B,Hq,Sk,D]; the host sums the
  // GQA groups (rep partials -> one kv head) in one cheap fused pass
  ushort* dkb = dk + bh * Sk * (int64_t)D;
  ushort* dvb = dv + bh * Sk * (int64_t)D;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int kr = k0 + wave * 32 + mi * 16 + (lane >> 4) * 4 + j;
      if (kr >= Sk) continue;
#pragma unroll
      for (int nd = 0; nd < D / 16; ++nd) {
        dkb[(int64_t)kr * D + nd * 16 + (lane & 15)] = af2bf(dkacc[mi][nd][j]);
        dvb[(int64_t)kr * D + nd * 16 + (lane & 15)] = af2bf(dvacc[mi][nd][j]);
      }
    }
  }
}

// pass B: dq. Block owns queries [q0, q0+128); wave w owns rows [w*32, +32).
// Q and dO rows live in registers as A-fragments; K,V tiles of 64 keys are
// staged per inner step (K in both layouts).
template <int D>
__global__ __launch_bounds__(256, 2) void fa_bwd_dq_kernel(
    const abf16* __restrict__ q, const abf16* __restrict__ k,
    const abf16* __restrict__ v, const abf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ drow,
    ushort* __restrict__ dq, int Sq, int Sk, int past, int causal, float scale,
    int Hq, int Hkv, Str3 sQ, Str3 sK, Str3 sV, Str3 sDo) {
  constexpr int BM = 128, BN = 64;
  constexpr int KP = D + 8, NP = BN + 8;
  extern __shared__ char smem[];
  abf16* Ks = reinterpret_cast<abf16*>(smem);  // [BN][KP]
  abf16* KTs = Ks + BN * KP;                   // [D][NP]
  abf16* Vs = KTs + D * NP;                    // [BN][KP]
  abf16* dSs = Vs + BN * KP;                   // [BM][NP]

  const int tid = threadIdx.x, wave = tid >> 6, lane = tid & 63;
  const int q0 = blockIdx.x * BM;
  const int64_t bh = blockIdx.y;
  const int64_t b = bh / Hq, h = bh % Hq, hk = h / (Hq / Hkv);
  const abf16* qb = q + b * sQ.b + h * sQ.h;
  const abf16* kb = k + b * sK.b + hk * sK.h;
  const abf16* vb = v + b * sV.b + hk * sV.h;
  const abf16* dob = dout + b * sDo.b + h * sDo.h;

  bf16x8a qf[2][D / 32], dof[2][D / 32];
  float l2r[2][4], drr[2][4];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
    int qr = q0 + wave * 32 + mi * 16 + (lane & 15);
    if (qr >= Sq) qr = Sq - 1;
#pragma unroll
    for (int kc = 0; kc < D / 32; ++kc) {
      qf[mi][kc] = *reinterpret_cast<const bf16x8a*>(qb + (int64_t)qr * sQ.s + kc * 32 + (lane >> 4) * 8);
      dof[mi][kc] = *reinterpret_cast<const bf16x8a*>(dob + (int64_t)qr * sDo.s + kc * 32 + (lane >> 4) * 8);
    }
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      int rr = q0 + wave * 32 + mi * 16 + (lane >> 4) * 4 + j;
      if (rr >= Sq) rr = Sq - 1;
      l2r[mi][j] = lse[bh * Sq + rr] * kLog2e;
      drr[mi][j] = drow[bh * Sq + rr];
    }
  }
  f32x4a dqacc[2][D / 16];
#pragma unroll
  for (int mi = 0; mi < 2; ++mi)
#pragma unroll
    for (int nd = 0; nd < D / 16; ++nd) dqacc[mi][nd] = f32x4a{};

  const float c = scale * kLog2e;
  const int k_hi = causal ? min(Sk, past + q0 + BM) : Sk;

  for (int k0 = 0; k0 < k_hi; k0 += BN) {
    {
      constexpr int pair_chunks = (BN / 2) * (D / 8);
#pragma unroll
      for (int it = 0; it < pair_chunks / 256 + (pair_chunks % 256 != 0); ++it) {
        const int cid = tid + it * 256;
        if (pair_chunks % 256 != 0 && cid >= pair_chunks) break;
        const int r0 = (cid / (D / 8)) * 2;
        const int col8 = (cid % (D / 8)) * 8;
        int64_t kr0 = k0 + r0, kr1 = k0 + r0 + 1;
        if (kr0 >= Sk) kr0 = Sk - 1;
        if (kr1 >= Sk) kr1 = Sk - 1;
        bf16x8a ka = *reinterpret_cast<const bf16x8a*>(kb + kr0 * sK.s + col8);
        bf16x8a kc8 = *reinterpret_cast<const bf16x8a*>(kb + kr1 * sK.s + col8);
        *reinterpret_cast<bf16x8a*>(Ks + r0 * KP + col8) = ka;
        *reinterpret_cast<bf16x8a*>(Ks + (r0 + 1) * KP + col8) = kc8;
        *reinterpret_cast<bf16x8a*>(Vs + r0 * KP + col8) =
            *reinterpret_cast<const bf16x8a*>(vb + kr0 * sV.s + col8);
        *reinterpret_cast<bf16x8a*>(Vs + (r0 + 1) * KP + col8) =
            *reinterpret_cast<const bf16x8a*>(vb + kr1 * sV.s + col8);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          unsigned pk = bfbits(ka[j]) | (bfbits(kc8[j]) << 16);
          *reinterpret_cast<unsigned*>(KTs + (col8 + j) * NP + r0) = pk;
        }
      }
    }
    __syncthreads();

    f32x4a st[2][BN / 16], dpt[2][BN / 16];
#pragma unroll
    for (int mi = 0; mi < 2; ++mi)
#pragma unroll
      for (int ni = 0; ni < BN / 16; ++ni) { st[mi][ni] = f32x4a{}; dpt[mi][ni] = f32x4a{}; }
#pragma unroll
    for (int ni = 0; ni < BN / 16; ++ni) {
#pragma unroll
      for (int kc = 0; kc < D / 32; ++kc) {
        bf16x8a bk = *reinterpret_cast<const bf16x8a*>(
            Ks + (ni * 16 + (lane & 15)) * KP + kc * 32 + (lane >> 4) * 8);
        bf16x8a bv = *reinterpret_cast<const bf16x8a*>(
            Vs + (ni * 16 + (lane & 15)) * KP + kc * 32 + (lane >> 4) * 8);
#pragma unroll
        for (int mi = 0; mi < 2; ++mi) {
          st[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[mi][kc], bk, st[mi][ni], 0, 0, 0);
          dpt[mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(dof[mi][kc], bv, dpt[mi][ni], 0, 0, 0);
        }
      }
    }

    // dS = P ⊙ (dP − Drow)·scale, stage as A-frags (wave-private rows)
#pragma unroll
    for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
      for (int ni = 0; ni < BN / 16; ++ni) {
        const int kc = k0 + ni * 16 + (lane & 15);
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          const int qr = q0 + wave * 32 + mi * 16 + (lane >> 4) * 4 + j;
          const bool dead = kc >= Sk || (causal && kc > past + qr);
          const float p = dead ? 0.f : fast_exp2(st[mi][ni][j] * c - l2r[mi][j]);
          const float ds = p * (dpt[mi][ni][j] - drr[mi][j]) * scale;
          ushort b = af2bf(ds);
          dSs[(wave * 32 + mi * 16 + (lane >> 4) * 4 + j) * NP + ni * 16 + (lane & 15)] =
              *reinterpret_cast<abf16*>(&b);
        }
      }
    }
    // dq += dS K (B-frags from K^T; contraction = keys)
#pragma unroll
    for (int kc = 0; kc < BN / 32; ++kc) {
      bf16x8a sf[2];
#pragma unroll
      for (int mi = 0; mi < 2; ++mi)
        sf[mi] = *reinterpret_cast<const bf16x8a*>(
            dSs + (wave * 32 + mi * 16 + (lane & 15)) * NP + kc * 32 + (lane >> 4) * 8);
#pragma unroll
      for (int nd = 0; nd < D / 16; ++nd) {
        bf16x8a bf = *reinterpret_cast<const bf16x8a*>(
            KTs + (nd * 16 + (lane & 15)) * NP + kc * 32 + (lane >> 4) * 8);
#pragma unroll
        for (int mi = 0; mi < 2; ++mi)
          dqacc[mi][nd] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(sf[mi], bf, dqacc[mi][nd], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  ushort* dqb = dq + bh * Sq * (int64_t)D;
#pragma unroll
  for (int mi = 0; mi < 2; ++mi) {
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      const int qr = q0 + wave * 32 + mi * 16 + (lane >> 4) * 4 + j;
      if (qr >= Sq) continue;
#pragma unroll
      for (int nd = 0; nd < D / 16; ++nd)
        dqb[(int64_t)qr * D + nd * 16 + (lane & 15)] = af2bf(dqacc[mi][nd][j]);
    }
  }
}

template __global__ void fa_bwd_dkdv_kernel<64, 32>(const abf16*, const abf16*, const abf16*,
                                                    const abf16*, const float*, const float*,
                                                    ushort*, ushort*, int, int, int, int, float,
                                                    int, int, Str3, Str3, Str3, Str3);
template __global__ void fa_bwd_dkdv_kernel<128, 32>(const abf16*, const abf16*, const abf16*,
                                                     const abf16*, const float*, const float*,
                                                     ushort*, ushort*, int, int, int, int, float,
                                                     int, int, Str3, Str3, Str3, Str3);

// Swapped-orientation dv / dk (D=128), split into two kernels so each
// stays inside the 256-VGPR budget at 8 waves (the fused variant needs
// K+V residency + two 64-reg accumulators + two 32-reg score fragments
// at once). Each wave owns 32 KEY rows lane-resident (K rows for dv,
// K+V rows for dk); 64-row query tiles stream through LDS. The score
// orientation is mfma(Q_lds, K_reg) -> C[q][k] with col = k (the lane's
// resident key), so the P^T/dS^T relayout feeds the dv/dk MFMAs with the
// same pack+permlane32_swap recipe as the forward.

template <int D>
__global__ __launch_bounds__(512, 1) void fa_bwd_dv_swapped_kernel(
    const abf16* __restrict__ q, const abf16* __restrict__ k,
    const abf16* __restrict__ dout, const float* __restrict__ lse,
    ushort* __restrict__ dv, int Sq, int Sk, int past, int causal, float scale,
    int Hq, int Hkv, Str3 sQ, Str3 sK, Str3 sDo) {
  static_assert(D == 128);
  constexpr int BK = 256, BQ = 64;
  constexpr int KP = D + 8;
  constexpr int NP = BQ + 8;
  extern __shared__ char smem[];
  abf16* Qs = reinterpret_cast<abf16*>(smem);  // [BQ][KP] row-major
  abf16* DOTs = Qs + BQ * KP;                  // [D][NP] dOut^T
  float* lse_s = reinterpret_cast<float*>(DOTs + D * NP);  // [BQ]

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;
  // causal: blocks over LOW keys see the most q-tiles -> launch them first
  const int k0 = (causal ? (int)blockIdx.x : (int)blockIdx.x) * BK;
  const int64_t bh = blockIdx.y;
  const int64_t b = bh / Hq, h = bh % Hq, hk = h / (Hq / Hkv);
  const abf16* qb = q + b * sQ.b + h * sQ.h;
  const abf16* kb = k + b * sK.b + hk * sK.h;
  const abf16* dob = dout + b * sDo.b + h * sDo.h;

  int kr_mine = k0 + wave * 32 + col;
  if (kr_mine >= Sk) kr_mine = Sk - 1;  // clamp (stores guarded)
  bf16x8a kfr[D / 16];
#pragma unroll
  for (int t = 0; t < D / 16; ++t)
    kfr[t] = *reinterpret_cast<const bf16x8a*>(kb + (int64_t)kr_mine * sK.s + t * 16 + hi * 8);

  f32x16a dvacc[D / 32] = {};
  const float c = scale * kLog2e;
  const int kabs = k0 + wave * 32 + col;  // this lane's key (column)
  // causal: key kb attends only q >= kb - past
  const int q_lo_blk = causal ? max(0, ((k0 - past) / BQ) * BQ) : 0;
  const int q_lo_wave = causal ? (k0 + wave * 32 - past) : 0;

  for (int qt0 = q_lo_blk; qt0 < Sq; qt0 += BQ) {
    {  // stage Q rows, dOut^T, lse tile
#pragma unroll
      for (int it = 0; it < BQ * (D / 8) / 512; ++it) {
        const int cid = tid + it * 512;
        const int r = cid / (D / 8);
        const int col8 = (cid % (D / 8)) * 8;
        int64_t qrow = qt0 + r;
        if (qrow >= Sq) qrow = Sq - 1;
        *reinterpret_cast<bf16x8a*>(Qs + r * KP + col8) =
            *reinterpret_cast<const bf16x8a*>(qb + qrow * sQ.s + col8);
      }
      const int r0 = (tid / (D / 8)) * 2;
      const int col8 = (tid % (D / 8)) * 8;
      int64_t qr0 = qt0 + r0, qr1 = qt0 + r0 + 1;
      if (qr0 >= Sq) qr0 = Sq - 1;
      if (qr1 >= Sq) qr1 = Sq - 1;
      bf16x8a d0 = *reinterpret_cast<const bf16x8a*>(dob + qr0 * sDo.s + col8);
      bf16x8a d1 = *reinterpret_cast<const bf16x8a*>(dob + qr1 * sDo.s + col8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const unsigned pk = bfbits(d0[j]) | (bfbits(d1[j]) << 16);
        *reinterpret_cast<unsigned*>(DOTs + (col8 + j) * NP + r0) = pk;
      }
      if (tid < BQ) lse_s[tid] = lse[bh * Sq + min(qt0 + tid, Sq - 1)];
    }
    __syncthreads();

    if (qt0 + BQ > q_lo_wave) {  // tiles fully below the horizon: barriers only
      f32x16a sacc[2] = {f32x16a{}, f32x16a{}};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int qt = 0; qt < 2; ++qt)
#pragma unroll
        for (int t = 0; t < D / 16; ++t) {
          bf16x8a qff = *reinterpret_cast<const bf16x8a*>(
              Qs + (qt * 32 + col) * KP + t * 16 + hi * 8);
          sacc[qt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qff, kfr[t], sacc[qt], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);

      // P^T = exp2(S*c - lse[q]); zero masked (causal below-horizon, q tail)
      const bool edge = (qt0 + BQ > Sq) || (causal && qt0 < q_lo_wave + 32);
#pragma unroll
      for (int qt = 0; qt < 2; ++qt) {
        const int qbase = qt0 + qt * 32 + 4 * hi;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qrow = qt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          float pv = fast_exp2(sacc[qt][r] * c - lse_s[qrow] * kLog2e);
          if (edge) {
            const int qa = qbase + (r & 3) + 8 * (r >> 2);
            if (qa >= Sq || (causal && kabs > past + qa)) pv = 0.f;
          }
          sacc[qt][r] = pv;
        }
      }

      unsigned pa[4][4] __attribute__((aligned(16)));
#pragma unroll
      for (int qs = 0; qs < 4; ++qs) {
        const f32x16a& pp = sacc[qs >> 1];
        const int base = 8 * (qs & 1);
        uint2a r01 = __builtin_amdgcn_permlane32_swap(
            pack_bf16_pair(pp[base + 0], pp[base + 1]),
            pack_bf16_pair(pp[base + 4], pp[base + 5]), false, false);
        uint2a r23 = __builtin_amdgcn_permlane32_swap(
            pack_bf16_pair(pp[base + 2], pp[base + 3]),
            pack_bf16_pair(pp[base + 6], pp[base + 7]), false, false);
        pa[qs][0] = r01.x;
        pa[qs][1] = r23.x;
        pa[qs][2] = r01.y;
        pa[qs][3] = r23.y;
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
        for (int qs = 0; qs < 4; ++qs) {
          bf16x8a dof = *reinterpret_cast<const bf16x8a*>(
              DOTs + (dt * 32 + col) * NP + qs * 16 + hi * 8);
          dvacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<bf16x8a*>(pa[qs]), dof, dvacc[dt], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
  }

  ushort* dvb = dv + bh * Sk * (int64_t)D;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kr = k0 + wave * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
    if (kr >= Sk) continue;
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt)
      dvb[(int64_t)kr * D + dt * 32 + col] = af2bf(dvacc[dt][r]);
  }
}

template <int D>
__global__ __launch_bounds__(512, 1) void fa_bwd_dk_swapped_kernel(
    const abf16* __restrict__ q, const abf16* __restrict__ k,
    const abf16* __restrict__ v, const abf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ drow,
    ushort* __restrict__ dk, int Sq, int Sk, int past, int causal, float scale,
    int Hq, int Hkv, Str3 sQ, Str3 sK, Str3 sV, Str3 sDo) {
  static_assert(D == 128);
  constexpr int BK = 256, BQ = 64;
  constexpr int KP = D + 8;
  constexpr int NP = BQ + 8;
  extern __shared__ char smem[];
  abf16* Qs = reinterpret_cast<abf16*>(smem);  // [BQ][KP]
  abf16* DOs = Qs + BQ * KP;                   // [BQ][KP] dOut rows
  abf16* QTs = DOs + BQ * KP;                  // [D][NP]  Q^T
  float* lse_s = reinterpret_cast<float*>(QTs + D * NP);   // [BQ]
  float* drow_s = lse_s + BQ;                              // [BQ]

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;
  const int k0 = (int)blockIdx.x * BK;
  const int64_t bh = blockIdx.y;
  const int64_t b = bh / Hq, h = bh % Hq, hk = h / (Hq / Hkv);
  const abf16* qb = q + b * sQ.b + h * sQ.h;
  const abf16* kb = k + b * sK.b + hk * sK.h;
  const abf16* vb = v + b * sV.b + hk * sV.h;
  const abf16* dob = dout + b * sDo.b + h * sDo.h;

  int kr_mine = k0 + wave * 32 + col;
  if (kr_mine >= Sk) kr_mine = Sk - 1;
  bf16x8a kfr[D / 16], vfr[D / 16];
#pragma unroll
  for (int t = 0; t < D / 16; ++t) {
    kfr[t] = *reinterpret_cast<const bf16x8a*>(kb + (int64_t)kr_mine * sK.s + t * 16 + hi * 8);
    vfr[t] = *reinterpret_cast<const bf16x8a*>(vb + (int64_t)kr_mine * sV.s + t * 16 + hi * 8);
  }

  f32x16a dkacc[D / 32] = {};
  const float c = scale * kLog2e;
  const int kabs = k0 + wave * 32 + col;
  const int q_lo_blk = causal ? max(0, ((k0 - past) / BQ) * BQ) : 0;
  const int q_lo_wave = causal ? (k0 + wave * 32 - past) : 0;

  for (int qt0 = q_lo_blk; qt0 < Sq; qt0 += BQ) {
    {  // stage Q rows, dOut rows, Q^T, lse+drow tiles
#pragma unroll
      for (int it = 0; it < 2 * BQ * (D / 8) / 512; ++it) {
        const int cid = tid + it * 512;
        const int r = (cid / (D / 8)) % BQ;
        const bool is_do = cid >= BQ * (D / 8);
        const int col8 = (cid % (D / 8)) * 8;
        int64_t qrow = qt0 + r;
        if (qrow >= Sq) qrow = Sq - 1;
        if (is_do)
          *reinterpret_cast<bf16x8a*>(DOs + r * KP + col8) =
              *reinterpret_cast<const bf16x8a*>(dob + qrow * sDo.s + col8);
        else
          *reinterpret_cast<bf16x8a*>(Qs + r * KP + col8) =
              *reinterpret_cast<const bf16x8a*>(qb + qrow * sQ.s + col8);
      }
      const int r0 = (tid / (D / 8)) * 2;
      const int col8 = (tid % (D / 8)) * 8;
      int64_t qr0 = qt0 + r0, qr1 = qt0 + r0 + 1;
      if (qr0 >= Sq) qr0 = Sq - 1;
      if (qr1 >= Sq) qr1 = Sq - 1;
      bf16x8a a0 = *reinterpret_cast<const bf16x8a*>(qb + qr0 * sQ.s + col8);
      bf16x8a a1 = *reinterpret_cast<const bf16x8a*>(qb + qr1 * sQ.s + col8);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        const unsigned pk = bfbits(a0[j]) | (bfbits(a1[j]) << 16);
        *reinterpret_cast<unsigned*>(QTs + (col8 + j) * NP + r0) = pk;
      }
      if (tid < BQ) {
        lse_s[tid] = lse[bh * Sq + min(qt0 + tid, Sq - 1)];
        drow_s[tid] = drow[bh * Sq + min(qt0 + tid, Sq - 1)];
      }
    }
    __syncthreads();

    if (qt0 + BQ > q_lo_wave) {
      f32x16a sacc[2] = {f32x16a{}, f32x16a{}};
      f32x16a dpacc[2] = {f32x16a{}, f32x16a{}};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int qt = 0; qt < 2; ++qt)
#pragma unroll
        for (int t = 0; t < D / 16; ++t) {
          bf16x8a qff = *reinterpret_cast<const bf16x8a*>(
              Qs + (qt * 32 + col) * KP + t * 16 + hi * 8);
          sacc[qt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qff, kfr[t], sacc[qt], 0, 0, 0);
          bf16x8a dff = *reinterpret_cast<const bf16x8a*>(
              DOs + (qt * 32 + col) * KP + t * 16 + hi * 8);
          dpacc[qt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dff, vfr[t], dpacc[qt], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);

      // dS^T = P^T (dP^T - Drow[q]) scale (masked -> 0)
      const bool edge = (qt0 + BQ > Sq) || (causal && qt0 < q_lo_wave + 32);
#pragma unroll
      for (int qt = 0; qt < 2; ++qt) {
        const int qbase = qt0 + qt * 32 + 4 * hi;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qrow = qt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
          float pv = fast_exp2(sacc[qt][r] * c - lse_s[qrow] * kLog2e);
          if (edge) {
            const int qa = qbase + (r & 3) + 8 * (r >> 2);
            if (qa >= Sq || (causal && kabs > past + qa)) pv = 0.f;
          }
          sacc[qt][r] = pv * (dpacc[qt][r] - drow_s[qrow]) * scale;
        }
      }

      unsigned pa[4][4] __attribute__((aligned(16)));
#pragma unroll
      for (int qs = 0; qs < 4; ++qs) {
        const f32x16a& pp = sacc[qs >> 1];
        const int base = 8 * (qs & 1);
        uint2a r01 = __builtin_amdgcn_permlane32_swap(
            pack_bf16_pair(pp[base + 0], pp[base + 1]),
            pack_bf16_pair(pp[base + 4], pp[base + 5]), false, false);
        uint2a r23 = __builtin_amdgcn_permlane32_swap(
            pack_bf16_pair(pp[base + 2], pp[base + 3]),
            pack_bf16_pair(pp[base + 6], pp[base + 7]), false, false);
        pa[qs][0] = r01.x;
        pa[qs][1] = r23.x;
        pa[qs][2] = r01.y;
        pa[qs][3] = r23.y;
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
        for (int qs = 0; qs < 4; ++qs) {
          bf16x8a qtf = *reinterpret_cast<const bf16x8a*>(
              QTs + (dt * 32 + col) * NP + qs * 16 + hi * 8);
          dkacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<bf16x8a*>(pa[qs]), qtf, dkacc[dt], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
  }

  ushort* dkb = dk + bh * Sk * (int64_t)D;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kr = k0 + wave * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
    if (kr >= Sk) continue;
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt)
      dkb[(int64_t)kr * D + dt * 32 + col] = af2bf(dkacc[dt][r]);
  }
}

template __global__ void fa_bwd_dv_swapped_kernel<128>(const abf16*, const abf16*, const abf16*,
                                                       const float*, ushort*, int, int, int, int,
                                                       float, int, int, Str3, Str3, Str3);
template __global__ void fa_bwd_dk_swapped_kernel<128>(const abf16*, const abf16*, const abf16*,
                                                       const abf16*, const float*, const float*,
                                                       ushort*, int, int, int, int, float,
                                                       int, int, Str3, Str3, Str3, Str3);

// Fused swapped dk+dv (D=128, BQ=32): one pass recomputes S once for both
// grads. Register liveness is sequenced so the P fragments are consumed by
// the dv MFMAs BEFORE dP materializes (peak stays under the 256-VGPR cap):
//   S -> P(in place) -> pa_p -> dv += pa_p x dOut^T -> dP -> dS(in place)
//   -> pa_ds -> dk += pa_ds x Q^T
template <int D>
__global__ __launch_bounds__(512, 1) void fa_bwd_dkdv_swapped_kernel(
    const abf16* __restrict__ q, const abf16* __restrict__ k,
    const abf16* __restrict__ v, const abf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ drow,
    ushort* __restrict__ dk, ushort* __restrict__ dv,
    int Sq, int Sk, int past, int causal, float scale,
    int Hq, int Hkv, Str3 sQ, Str3 sK, Str3 sV, Str3 sDo) {
  static_assert(D == 64 || D == 128);
  constexpr int BK = 256, BQ = 32;
  constexpr int KP = D + 8;
  constexpr int NP = BQ + 8;
  extern __shared__ char smem[];
  abf16* Qs = reinterpret_cast<abf16*>(smem);  // [BQ][KP]
  abf16* DOs = Qs + BQ * KP;                   // [BQ][KP]
  abf16* QTs = DOs + BQ * KP;                  // Q subtiled [q/4][d/16][4][16]
  abf16* DOTs = QTs + BQ * D;                  // dOut subtiled likewise
  float* lse_s = reinterpret_cast<float*>(DOTs + BQ * D);  // [BQ] (base-2)
  float* drow_s = lse_s + BQ;                              // [BQ]

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;
  const int k0 = (int)blockIdx.x * BK;  // low keys = most q-tiles: heavy-first
  const int64_t bh = blockIdx.y;
  const int64_t b = bh / Hq, h = bh % Hq, hk = h / (Hq / Hkv);
  const abf16* qb = q + b * sQ.b + h * sQ.h;
  const abf16* kb = k + b * sK.b + hk * sK.h;
  const abf16* vb = v + b * sV.b + hk * sV.h;
  const abf16* dob = dout + b * sDo.b + h * sDo.h;

  int kr_mine = k0 + wave * 32 + col;
  if (kr_mine >= Sk) kr_mine = Sk - 1;
  bf16x8a kfr[D / 16], vfr[D / 16];
#pragma unroll
  for (int t = 0; t < D / 16; ++t) {
    kfr[t] = *reinterpret_cast<const bf16x8a*>(kb + (int64_t)kr_mine * sK.s + t * 16 + hi * 8);
    vfr[t] = *reinterpret_cast<const bf16x8a*>(vb + (int64_t)kr_mine * sV.s + t * 16 + hi * 8);
  }

  f32x16a dkacc[D / 32] = {}, dvacc[D / 32] = {};
  const float c = scale * kLog2e;
  const int kabs = k0 + wave * 32 + col;
  const int q_lo_blk = causal ? max(0, ((k0 - past) / BQ) * BQ) : 0;
  const int q_lo_wave = causal ? (k0 + wave * 32 - past) : 0;

  for (int qt0 = q_lo_blk; qt0 < Sq; qt0 += BQ) {
    {  // stage: Q rows + dOut rows (BQ*(D/8) = 512 chunks each)
#pragma unroll
      for (int it = 0; it < 2; ++it) {
        const bool second = it == 1;
        const int r = tid / (D / 8);
        const int col8 = (tid % (D / 8)) * 8;
        if (r >= BQ) break;  // D=64: BQ*(D/8)=256 chunks per tensor
        int64_t qrow = qt0 + r;
        if (qrow >= Sq) qrow = Sq - 1;
        if (second)
          *reinterpret_cast<bf16x8a*>(DOs + r * KP + col8) =
              *reinterpret_cast<const bf16x8a*>(dob + qrow * sDo.s + col8);
        else
          *reinterpret_cast<bf16x8a*>(Qs + r * KP + col8) =
              *reinterpret_cast<const bf16x8a*>(qb + qrow * sQ.s + col8);
      }
      // transposes: threads 0-255 pair-pack Q^T, 256-511 dOut^T
      const bool tsecond = tid >= 256;
      const int tp = tid & 255;
      const int r0 = (tp / (D / 8)) * 2;
      const int col8t = (tp % (D / 8)) * 8;
      if (r0 < BQ) {  // D=64: (BQ/2)*(D/8)=128 pair-chunks per tensor
        int64_t qr0 = qt0 + r0, qr1 = qt0 + r0 + 1;
        if (qr0 >= Sq) qr0 = Sq - 1;
        if (qr1 >= Sq) qr1 = Sq - 1;
        const abf16* src = tsecond ? dob : qb;
        const int64_t ss = tsecond ? sDo.s : sQ.s;
        abf16* dst = tsecond ? DOTs : QTs;
        bf16x8a a0 = *reinterpret_cast<const bf16x8a*>(src + qr0 * ss + col8t);
        bf16x8a a1 = *reinterpret_cast<const bf16x8a*>(src + qr1 * ss + col8t);
        const int dbq = (col8t >> 4) * 64 + (col8t & 15);
        *reinterpret_cast<bf16x8a*>(dst + ((r0 >> 2) * (D / 16)) * 64 + dbq + (r0 & 3) * 16) = a0;
        *reinterpret_cast<bf16x8a*>(dst + (((r0 + 1) >> 2) * (D / 16)) * 64 + dbq +
                                    ((r0 + 1) & 3) * 16) = a1;
      }
      if (tid < BQ) {
        lse_s[tid] = lse[bh * Sq + min(qt0 + tid, Sq - 1)] * kLog2e;
        drow_s[tid] = drow[bh * Sq + min(qt0 + tid, Sq - 1)];
      }
    }
    __syncthreads();

    if (qt0 + BQ > q_lo_wave) {
      // --- S^T = mfma(Q, K): col = k, rows = q ---
      f32x16a sacc = {};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int t = 0; t < D / 16; ++t) {
        bf16x8a qff = *reinterpret_cast<const bf16x8a*>(Qs + col * KP + t * 16 + hi * 8);
        sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qff, kfr[t], sacc, 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);

      // P^T (masked -> 0)
      const bool edge = (qt0 + BQ > Sq) || (causal && qt0 < q_lo_wave + 32);
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
        float pv = fast_exp2(sacc[r] * c - lse_s[qrow]);
        if (edge) {
          const int qa = qt0 + qrow;
          if (qa >= Sq || (causal && kabs > past + qa)) pv = 0.f;
        }
        sacc[r] = pv;
      }
      unsigned pa[2][4] __attribute__((aligned(16)));
#pragma unroll
      for (int qs = 0; qs < 2; ++qs) {
        const int base = 8 * qs;
        uint2a r01 = __builtin_amdgcn_permlane32_swap(
            pack_bf16_pair(sacc[base + 0], sacc[base + 1]),
            pack_bf16_pair(sacc[base + 4], sacc[base + 5]), false, false);
        uint2a r23 = __builtin_amdgcn_permlane32_swap(
            pack_bf16_pair(sacc[base + 2], sacc[base + 3]),
            pack_bf16_pair(sacc[base + 6], sacc[base + 7]), false, false);
        pa[qs][0] = r01.x;
        pa[qs][1] = r23.x;
        pa[qs][2] = r01.y;
        pa[qs][3] = r23.y;
      }
      // dv += P^T x dOut^T (consumes pa before dP lives)
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
        for (int qs = 0; qs < 2; ++qs) {
          const unsigned a0 =
              64u * ((4 * qs + 2 * hi) * (D / 16) + 2 * dt + ((lane >> 4) & 1)) +
              (lane & 15) * 4;
          bf16x4t lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (__attribute__((address_space(3))) bf16x4t*)(DOTs + a0));
          bf16x4t hh = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (__attribute__((address_space(3))) bf16x4t*)(DOTs + a0 + 64 * (D / 16)));
          bf16x8a dof = bf16x8a{lo[0], lo[1], lo[2], lo[3], hh[0], hh[1], hh[2], hh[3]};
          dvacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<bf16x8a*>(pa[qs]), dof, dvacc[dt], 0, 0, 0);
        }
      // dP^T = mfma(dOut, V)
      f32x16a dpacc = {};
#pragma unroll
      for (int t = 0; t < D / 16; ++t) {
        bf16x8a dff = *reinterpret_cast<const bf16x8a*>(DOs + col * KP + t * 16 + hi * 8);
        dpacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dff, vfr[t], dpacc, 0, 0, 0);
      }
      __builtin_amdgcn_s_setprio(0);
      // dS^T = P (dP - Drow) scale
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qrow = (r & 3) + 8 * (r >> 2) + 4 * hi;
        sacc[r] = sacc[r] * (dpacc[r] - drow_s[qrow]) * scale;
      }
#pragma unroll
      for (int qs = 0; qs < 2; ++qs) {
        const int base = 8 * qs;
        uint2a r01 = __builtin_amdgcn_permlane32_swap(
            pack_bf16_pair(sacc[base + 0], sacc[base + 1]),
            pack_bf16_pair(sacc[base + 4], sacc[base + 5]), false, false);
        uint2a r23 = __builtin_amdgcn_permlane32_swap(
            pack_bf16_pair(sacc[base + 2], sacc[base + 3]),
            pack_bf16_pair(sacc[base + 6], sacc[base + 7]), false, false);
        pa[qs][0] = r01.x;
        pa[qs][1] = r23.x;
        pa[qs][2] = r01.y;
        pa[qs][3] = r23.y;
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
        for (int qs = 0; qs < 2; ++qs) {
          const unsigned a0 =
              64u * ((4 * qs + 2 * hi) * (D / 16) + 2 * dt + ((lane >> 4) & 1)) +
              (lane & 15) * 4;
          bf16x4t lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (__attribute__((address_space(3))) bf16x4t*)(QTs + a0));
          bf16x4t hh = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (__attribute__((address_space(3))) bf16x4t*)(QTs + a0 + 64 * (D / 16)));
          bf16x8a qtf = bf16x8a{lo[0], lo[1], lo[2], lo[3], hh[0], hh[1], hh[2], hh[3]};
          dkacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<bf16x8a*>(pa[qs]), qtf, dkacc[dt], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
  }

  ushort* dkb = dk + bh * Sk * (int64_t)D;
  ushort* dvb = dv + bh * Sk * (int64_t)D;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int kr = k0 + wave * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
    if (kr >= Sk) continue;
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt) {
      dkb[(int64_t)kr * D + dt * 32 + col] = af2bf(dkacc[dt][r]);
      dvb[(int64_t)kr * D + dt * 32 + col] = af2bf(dvacc[dt][r]);
    }
  }
}

template __global__ void fa_bwd_dkdv_swapped_kernel<128>(
    const abf16*, const abf16*, const abf16*, const abf16*, const float*, const float*,
    ushort*, ushort*, int, int, int, int, float, int, int, Str3, Str3, Str3, Str3);
template __global__ void fa_bwd_dkdv_swapped_kernel<64>(
    const abf16*, const abf16*, const abf16*, const abf16*, const float*, const float*,
    ushort*, ushort*, int, int, int, int, float, int, int, Str3, Str3, Str3, Str3);

// Swapped-orientation dq (D=128): same 8-wave 32x32 ladder as the forward.
// Per wave, 32 query rows are lane-resident (Q and dOut B-fragments, lse
// and Drow are per-lane scalars); K tiles of 64 stream through LDS as
// row-major A-sources (S^T = mfma(K,Q), dP^T = mfma(V,dOut)) plus a
// transposed K^T copy for the dq += dS^T-relayout x K^T MFMAs. No online
// softmax (lse recompute), so the per-tile VALU cost is one exp2 + the
// dS arithmetic per score.
template <int D>
__global__ __launch_bounds__(512, 1) void fa_bwd_dq_swapped_kernel(
    const abf16* __restrict__ q, const abf16* __restrict__ k,
    const abf16* __restrict__ v, const abf16* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ drow,
    ushort* __restrict__ dq, int Sq, int Sk, int past, int causal, float scale,
    int Hq, int Hkv, Str3 sQ, Str3 sK, Str3 sV, Str3 sDo) {
  static_assert(D == 64 || D == 128, "swapped dq supports head_dim 64/128");
  constexpr int BM = 256, BN = 64;
  constexpr int KP = D + 8;
  constexpr int NP = BN + 8;
  extern __shared__ char smem[];
  abf16* Ks = reinterpret_cast<abf16*>(smem);  // [BN][KP] row-major
  abf16* Vs = Ks + BN * KP;                    // [BN][KP] row-major
  abf16* KTs = Vs + BN * KP;                   // K subtiled [k/4][d/16][4][16] (tr_read)

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int hi = lane >> 5;
  const int col = lane & 31;
  const int q0 = (causal ? (int)(gridDim.x - 1 - blockIdx.x) : (int)blockIdx.x) * BM;
  const int64_t bh = blockIdx.y;
  const int64_t b = bh / Hq, h = bh % Hq, hk = h / (Hq / Hkv);
  const abf16* qb = q + b * sQ.b + h * sQ.h;
  const abf16* kb = k + b * sK.b + hk * sK.h;
  const abf16* vb = v + b * sV.b + hk * sV.h;
  const abf16* dob = dout + b * sDo.b + h * sDo.h;

  int qr_mine = q0 + wave * 32 + col;
  if (qr_mine >= Sq) qr_mine = Sq - 1;
  bf16x8a qf[D / 16], dof[D / 16];
#pragma unroll
  for (int t = 0; t < D / 16; ++t) {
    qf[t] = *reinterpret_cast<const bf16x8a*>(qb + (int64_t)qr_mine * sQ.s + t * 16 + hi * 8);
    dof[t] = *reinterpret_cast<const bf16x8a*>(dob + (int64_t)qr_mine * sDo.s + t * 16 + hi * 8);
  }
  // scale folds into the exp argument: dS = P*(dP-D)*scale
  //   = exp2(s*c - lse*log2e + log2(scale)) * (dP - D)
  const float lse2 = lse[bh * Sq + qr_mine] * kLog2e - __log2f(scale);
  const float dr_m = drow[bh * Sq + qr_mine];

  f32x16a dqacc[D / 32] = {};
  const float c = scale * kLog2e;
  const int qabs = q0 + wave * 32 + col;
  const int k_hi = causal ? min(Sk, past + q0 + BM) : Sk;
  const int k_hi_wave = causal ? min(Sk, past + q0 + wave * 32 + 32) : Sk;

  for (int kb0 = 0; kb0 < k_hi; kb0 += BN) {
    // --- stage K, V row-major and K^T (pair-packed) ---
    {
#pragma unroll
      for (int it = 0; it < 2 * BN * (D / 8) / 512; ++it) {
        const int cid = tid + it * 512;
        const int r = (cid / (D / 8)) % BN;
        const bool is_v = cid >= BN * (D / 8);
        const int col8 = (cid % (D / 8)) * 8;
        int64_t key = kb0 + r;
        if (key >= Sk) key = Sk - 1;  // clamp (P zeroed below)
        if (is_v)
          *reinterpret_cast<bf16x8a*>(Vs + r * KP + col8) =
              *reinterpret_cast<const bf16x8a*>(vb + key * sV.s + col8);
        else
          *reinterpret_cast<bf16x8a*>(Ks + r * KP + col8) =
              *reinterpret_cast<const bf16x8a*>(kb + key * sK.s + col8);
      }
      const int r0 = (tid / (D / 8)) * 2;
      const int col8 = (tid % (D / 8)) * 8;
      if (r0 < BN) {  // D=64: only BN/2*(D/8)=256 pair-chunks
        int64_t key0 = kb0 + r0, key1 = kb0 + r0 + 1;
        if (key0 >= Sk) key0 = Sk - 1;
        if (key1 >= Sk) key1 = Sk - 1;
        bf16x8a k0 = *reinterpret_cast<const bf16x8a*>(kb + key0 * sK.s + col8);
        bf16x8a k1 = *reinterpret_cast<const bf16x8a*>(kb + key1 * sK.s + col8);
        const int dbq = (col8 >> 4) * 64 + (col8 & 15);
        *reinterpret_cast<bf16x8a*>(KTs + ((r0 >> 2) * (D / 16)) * 64 + dbq + (r0 & 3) * 16) = k0;
        *reinterpret_cast<bf16x8a*>(KTs + (((r0 + 1) >> 2) * (D / 16)) * 64 + dbq +
                                    ((r0 + 1) & 3) * 16) = k1;
      }
    }
    __syncthreads();

    if (kb0 < k_hi_wave) {
      // --- S^T = mfma(K,Q), dP^T = mfma(V,dOut): col = q, rows = k ---
      f32x16a sacc[2] = {f32x16a{}, f32x16a{}};
      f32x16a dpacc[2] = {f32x16a{}, f32x16a{}};
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int kt = 0; kt < 2; ++kt)
#pragma unroll
        for (int t = 0; t < D / 16; ++t) {
          bf16x8a kf = *reinterpret_cast<const bf16x8a*>(
              Ks + (kt * 32 + col) * KP + t * 16 + hi * 8);
          sacc[kt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[t], sacc[kt], 0, 0, 0);
          bf16x8a vf = *reinterpret_cast<const bf16x8a*>(
              Vs + (kt * 32 + col) * KP + t * 16 + hi * 8);
          dpacc[kt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dof[t], dpacc[kt], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);

      // --- dS = P (dP - Drow) scale, with P zeroed on masked positions ---
      const bool edge = (kb0 + BN > Sk) || (causal && kb0 + BN > past + q0 + wave * 32);
#pragma unroll
      for (int kt = 0; kt < 2; ++kt) {
        const int kbase = kb0 + kt * 32 + 4 * hi;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float pv = fast_exp2(sacc[kt][r] * c - lse2);
          if (edge) {
            const int kabs = kbase + (r & 3) + 8 * (r >> 2);
            if (kabs >= Sk || (causal && kabs > past + qabs)) pv = 0.f;
          }
          sacc[kt][r] = pv * (dpacc[kt][r] - dr_m);
        }
      }

      // --- relayout dS into A-fragments, dq += dS^T-frags x K^T ---
      unsigned pa[4][4] __attribute__((aligned(16)));
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        const f32x16a& pp = sacc[ks >> 1];
        const int base = 8 * (ks & 1);
        uint2a r01 = __builtin_amdgcn_permlane32_swap(
            pack_bf16_pair(pp[base + 0], pp[base + 1]),
            pack_bf16_pair(pp[base + 4], pp[base + 5]), false, false);
        uint2a r23 = __builtin_amdgcn_permlane32_swap(
            pack_bf16_pair(pp[base + 2], pp[base + 3]),
            pack_bf16_pair(pp[base + 6], pp[base + 7]), false, false);
        pa[ks][0] = r01.x;
        pa[ks][1] = r23.x;
        pa[ks][2] = r01.y;
        pa[ks][3] = r23.y;
      }
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
        for (int ks = 0; ks < 4; ++ks) {
          const unsigned a0 =
              64u * ((4 * ks + 2 * hi) * (D / 16) + 2 * dt + ((lane >> 4) & 1)) +
              (lane & 15) * 4;
          bf16x4t lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (__attribute__((address_space(3))) bf16x4t*)(KTs + a0));
          bf16x4t hh = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (__attribute__((address_space(3))) bf16x4t*)(KTs + a0 + 64 * (D / 16)));
          bf16x8a ktf = bf16x8a{lo[0], lo[1], lo[2], lo[3], hh[0], hh[1], hh[2], hh[3]};
          dqacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              *reinterpret_cast<bf16x8a*>(pa[ks]), ktf, dqacc[dt], 0, 0, 0);
        }
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
  }

  ushort* dqb = dq + bh * Sq * (int64_t)D;
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int qr = q0 + wave * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
    if (qr >= Sq) continue;
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt)
      dqb[(int64_t)qr * D + dt * 32 + col] = af2bf(dqacc[dt][r]);
  }
}

template __global__ void fa_bwd_dq_swapped_kernel<128>(const abf16*, const abf16*, const abf16*,
                                                       const abf16*, const float*, const float*,
                                                       ushort*, int, int, int, int, float,
                                                       int, int, Str3, Str3, Str3, Str3);
template __global__ void fa_bwd_dq_swapped_kernel<64>(const abf16*, const abf16*, const abf16*,
                                                      const abf16*, const float*, const float*,
                                                      ushort*, int, int, int, int, float,
                                                      int, int, Str3, Str3, Str3, Str3);

template __global__ void fa_bwd_dq_kernel<64>(const abf16*, const abf16*, const abf16*,
                                              const abf16*, const float*, const float*,
                                              ushort*, int, int, int, int, float,
                                              int, int, Str3, Str3, Str3, Str3);
template __global__ void fa_bwd_dq_kernel<128>(const abf16*, const abf16*, const abf16*,
                                               const abf16*, const float*, const float*,
                                               ushort*, int, int, int, int, float,
                                               int, int, Str3, Str3, Str3, Str3);

namespace {

template <int D>
int fa_lds_bytes() {
  constexpr int BM = 128, BN = 128, NP = BN + 8, KP = D + 8;
  // P aliases the K tile when it fits (D=128); else it gets its own region
  const int elems = BN * KP + D * NP + (BM * NP <= BN * KP ? 0 : BM * NP);
  return elems * 2;
}

// 0 = legacy 4-wave 16x16 kernel, 1 = swapped-QK^T 8-wave 32x32 ladder
static int fa_fwd_impl() {
  static int impl = []() {
    const char* e = getenv("ACCELERATE_AMD_FA_FWD");
    if (e && strcmp(e, "legacy") == 0) return 0;
    if (e && strcmp(e, "swapped") == 0) return 1;
    if (e && strcmp(e, "swapped_tr") == 0) return 2;
    return 2;  // tr_read + dbuf + v_exp ladder: 926 TF/s fwd (profiles/fa_swapped_r2.md)
  }();
  return impl;
}

template <int D>
hipError_t launch_impl(const void* q, const void* k, const void* v, void* out, float* lse,
                       int64_t bh, int Sq, int Sk, int past, int causal, float scale,
                       int Hq, int Hkv, const Str3* strides, hipStream_t stream) {
  if (fa_fwd_impl() >= 1) {
    constexpr int lds_sw = 2 * (64 * (D + 8) + D * (64 + 8)) * 2;
    static bool attr_sw = false;
    auto kfn = fa_fwd_impl() == 2 ? &fa_fwd_swapped_kernel<D, true>
                                  : &fa_fwd_swapped_kernel<D, false>;
    if (!attr_sw) {
      hipError_t e = hipFuncSetAttribute(reinterpret_cast<const void*>(kfn),
                                         hipFuncAttributeMaxDynamicSharedMemorySize, lds_sw);
      if (e != hipSuccess) return e;
      attr_sw = true;
    }
    dim3 grid((Sq + 255) / 256, (unsigned)bh);
    hipLaunchKernelGGL(kfn, grid, dim3(512), lds_sw, stream,
                       reinterpret_cast<const abf16*>(q), reinterpret_cast<const abf16*>(k),
                       reinterpret_cast<const abf16*>(v), reinterpret_cast<ushort*>(out),
                       lse, Sq, Sk, past, causal, scale, Hq, Hkv,
                       strides[0], strides[1], strides[2], strides[3]);
    return hipGetLastError();
  }
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
                     lse, Sq, Sk, past, causal, scale, Hq, Hkv,
                     strides[0], strides[1], strides[2], strides[3]);
  return hipGetLastError();
}

}  // namespace

// raw launcher used by bindings.hip (template instantiations stay local).
// strides: q, k, v, out Str3 triples (elements).
hipError_t launch_fa_fwd(const void* q, const void* k, const void* v, void* out, float* lse,
                         int64_t batch_heads, int Sq, int Sk, int head_dim, int past,
                         int causal, float scale, int Hq, int Hkv, const Str3* strides,
                         hipStream_t stream) {
  if (head_dim == 64)
    return launch_impl<64>(q, k, v, out, lse, batch_heads, Sq, Sk, past, causal, scale,
                           Hq, Hkv, strides, stream);
  if (head_dim == 128)
    return launch_impl<128>(q, k, v, out, lse, batch_heads, Sq, Sk, past, causal, scale,
                            Hq, Hkv, strides, stream);
  return hipErrorInvalidValue;
}

namespace {

template <int D>
hipError_t launch_bwd_impl(const void* q, const void* k, const void* v, const void* dout,
                           const void* out, const float* lse, float* drow,
                           void* dq, void* dk, void* dv, int64_t bh, int Sq, int Sk,
                           int past, int causal, float scale,
                           int Hq, int Hkv, const Str3* strides, hipStream_t stream) {
  // BQ=32 q-tiles: ~47 KB LDS -> occupancy 2 blocks/CU; measured 16.15 ->
  // 11.08 ms/iter on the fused fwd+bwd microbench vs the BQ=64 variant
  constexpr int BQSEL = 32;
  const int QPp = BQSEL + 8;
  const int lds_dkdv = (2 * BQSEL * (D + 8) + 2 * D * QPp + 128 * QPp) * 2 + 2 * BQSEL * 4;
  const int lds_dq = (2 * 64 * (D + 8) + D * 72 + 128 * 72) * 2;
  static bool attr_set = false;
  if (!attr_set) {
    auto dkdv_attr_fn = &fa_bwd_dkdv_kernel<D, BQSEL>;
    hipError_t e = hipFuncSetAttribute(reinterpret_cast<const void*>(dkdv_attr_fn),
                                       hipFuncAttributeMaxDynamicSharedMemorySize, lds_dkdv);
    if (e != hipSuccess) return e;
    e = hipFuncSetAttribute(reinterpret_cast<const void*>(&fa_bwd_dq_kernel<D>),
                            hipFuncAttributeMaxDynamicSharedMemorySize, lds_dq);
    if (e != hipSuccess) return e;
    attr_set = true;
  }
  const int64_t n_rows = bh * Sq;
  const int64_t want_drow = (n_rows + 3) / 4;
  const int grid_drow = (int)(want_drow < 4096 ? want_drow : 4096);
  // strides[]: q, k, v, dout, out
  hipLaunchKernelGGL(fa_drow_kernel, dim3(grid_drow), dim3(256), 0, stream,
                     reinterpret_cast<const ushort*>(dout), reinterpret_cast<const ushort*>(out),
                     drow, n_rows, D, Hq, Sq, strides[3], strides[4]);
  // hipLaunchKernelGGL is a macro: template-ids with commas must go
  // through a function pointer
  // dkdv impl: fused swapped (default) | split | legacy (env)
  const char* dkdv_env = getenv("ACCELERATE_AMD_FA_BWD_DKDV");
  const int dkdv_mode = (dkdv_env && strcmp(dkdv_env, "legacy") == 0) ? 0
                        : (D == 128 && dkdv_env && strcmp(dkdv_env, "split") == 0) ? 1
                                                                                   : 2;
  if (dkdv_mode == 1) {
    constexpr int lds_dv = (64 * (D + 8) + D * (64 + 8)) * 2 + 64 * 4;
    constexpr int lds_dk = (2 * 64 * (D + 8) + D * (64 + 8)) * 2 + 2 * 64 * 4;
    static bool attr_sw2 = false;
    if (!attr_sw2) {
      hipError_t e =
          hipFuncSetAttribute(reinterpret_cast<const void*>(&fa_bwd_dv_swapped_kernel<128>),
                              hipFuncAttributeMaxDynamicSharedMemorySize, lds_dv);
      if (e != hipSuccess) return e;
      e = hipFuncSetAttribute(reinterpret_cast<const void*>(&fa_bwd_dk_swapped_kernel<128>),
                              hipFuncAttributeMaxDynamicSharedMemorySize, lds_dk);
      if (e != hipSuccess) return e;
      attr_sw2 = true;
    }
    dim3 gsw((Sk + 255) / 256, (unsigned)bh);
    hipLaunchKernelGGL(fa_bwd_dv_swapped_kernel<128>, gsw, dim3(512), lds_dv, stream,
                       reinterpret_cast<const abf16*>(q), reinterpret_cast<const abf16*>(k),
                       reinterpret_cast<const abf16*>(dout), lse,
                       reinterpret_cast<ushort*>(dv), Sq, Sk, past, causal, scale, Hq, Hkv,
                       strides[0], strides[1], strides[3]);
    hipLaunchKernelGGL(fa_bwd_dk_swapped_kernel<128>, gsw, dim3(512), lds_dk, stream,
                       reinterpret_cast<const abf16*>(q), reinterpret_cast<const abf16*>(k),
                       reinterpret_cast<const abf16*>(v), reinterpret_cast<const abf16*>(dout),
                       lse, drow, reinterpret_cast<ushort*>(dk), Sq, Sk, past, causal, scale,
                       Hq, Hkv, strides[0], strides[1], strides[2], strides[3]);
  } else if (dkdv_mode == 2) {
    constexpr int lds_f = (2 * 32 * (D + 8) + 2 * 32 * D) * 2 + 2 * 32 * 4;
    static bool attr_f = false;
    if (!attr_f) {
      hipError_t e =
          hipFuncSetAttribute(reinterpret_cast<const void*>(&fa_bwd_dkdv_swapped_kernel<D>),
                              hipFuncAttributeMaxDynamicSharedMemorySize, lds_f);
      if (e != hipSuccess) return e;
      attr_f = true;
    }
    hipLaunchKernelGGL(fa_bwd_dkdv_swapped_kernel<D>, dim3((Sk + 255) / 256, (unsigned)bh),
                       dim3(512), lds_f, stream,
                       reinterpret_cast<const abf16*>(q), reinterpret_cast<const abf16*>(k),
                       reinterpret_cast<const abf16*>(v), reinterpret_cast<const abf16*>(dout),
                       lse, drow, reinterpret_cast<ushort*>(dk), reinterpret_cast<ushort*>(dv),
                       Sq, Sk, past, causal, scale, Hq, Hkv,
                       strides[0], strides[1], strides[2], strides[3]);
  } else {
    auto dkdv_fn = &fa_bwd_dkdv_kernel<D, BQSEL>;
    hipLaunchKernelGGL(dkdv_fn, dim3((Sk + 127) / 128, (unsigned)bh), dim3(256),
                       lds_dkdv, stream,
                       reinterpret_cast<const abf16*>(q), reinterpret_cast<const abf16*>(k),
                       reinterpret_cast<const abf16*>(v), reinterpret_cast<const abf16*>(dout),
                       lse, drow, reinterpret_cast<ushort*>(dk), reinterpret_cast<ushort*>(dv),
                       Sq, Sk, past, causal, scale, Hq, Hkv,
                       strides[0], strides[1], strides[2], strides[3]);
  }
  const char* dq_env = getenv("ACCELERATE_AMD_FA_BWD_DQ");
  const bool dq_swapped = !(dq_env && strcmp(dq_env, "legacy") == 0);
  if (dq_swapped) {
    constexpr int lds_dq_sw = (2 * 64 * (D + 8) + 64 * D) * 2;
    static bool attr_sw = false;
    if (!attr_sw) {
      hipError_t e =
          hipFuncSetAttribute(reinterpret_cast<const void*>(&fa_bwd_dq_swapped_kernel<D>),
                              hipFuncAttributeMaxDynamicSharedMemorySize, lds_dq_sw);
      if (e != hipSuccess) return e;
      attr_sw = true;
    }
    hipLaunchKernelGGL(fa_bwd_dq_swapped_kernel<D>, dim3((Sq + 255) / 256, (unsigned)bh),
                       dim3(512), lds_dq_sw, stream,
                       reinterpret_cast<const abf16*>(q), reinterpret_cast<const abf16*>(k),
                       reinterpret_cast<const abf16*>(v), reinterpret_cast<const abf16*>(dout),
                       lse, drow, reinterpret_cast<ushort*>(dq), Sq, Sk, past, causal, scale,
                       Hq, Hkv, strides[0], strides[1], strides[2], strides[3]);
  } else {
    hipLaunchKernelGGL(fa_bwd_dq_kernel<D>, dim3((Sq + 127) / 128, (unsigned)bh), dim3(256),
                       lds_dq, stream,
                       reinterpret_cast<const abf16*>(q), reinterpret_cast<const abf16*>(k),
                       reinterpret_cast<const abf16*>(v), reinterpret_cast<const abf16*>(dout),
                       lse, drow, reinterpret_cast<ushort*>(dq), Sq, Sk, past, causal, scale,
                       Hq, Hkv, strides[0], strides[1], strides[2], strides[3]);
  }
  return hipGetLastError();
}

}  // namespace

hipError_t launch_fa_bwd(const void* q, const void* k, const void* v, const void* dout,
                         const void* out, const float* lse, float* drow,
                         void* dq, void* dk, void* dv, int64_t batch_heads, int Sq, int Sk,
                         int head_dim, int past, int causal, float scale,
                         int Hq, int Hkv, const Str3* strides, hipStream_t stream) {
  if (head_dim == 64)
    return launch_bwd_impl<64>(q, k, v, dout, out, lse, drow, dq, dk, dv, batch_heads,
                               Sq, Sk, past, causal, scale, Hq, Hkv, strides, stream);
  if (head_dim == 128)
    return launch_bwd_impl<128>(q, k, v, dout, out, lse, drow, dq, dk, dv, batch_heads,
                                Sq, Sk, past, causal, scale, Hq, Hkv, strides, stream);
  return hipErrorInvalidValue;
}
