// 8-phase 256x256-tile MFMA GEMM for gfx950: C[M,N] = A[M,K] @ B[N,K]^T
// (+bias), bf16 in, fp32 accumulate, bf16 out — the nn.Linear layout at
// large shapes (M,N % 256 == 0, K % 128 == 0; bindings fall back to the
// 128^2 kernel otherwise).
//
// This is the CDNA4 guide's "256^2 8-phase template" (§5) realized in
// plain HIP: 512 threads = 8 waves (2M x 4N), K-step 64, TWO K-tiles per
// iteration across 8 phases; each phase = {4-8 swizzled ds_read_b128 ||
// one half-tile global_load_lds prefetch} -> s_barrier -> lgkmcnt(0) ->
// setprio(1) -> 16 MFMA -> setprio(0) -> s_barrier, with COUNTED
// s_waitcnt vmcnt(4) only at phases 4/8 so prefetches stay in flight
// across barriers. LDS 128 KiB (2 dbuf x A,B x 2 halves x [128][64]).
// The XOR source-pre-swizzle (chunk ^= row&7 at 16 B granularity) makes
// every fragment ds_read 2-way-conflict-free: SQ_LDS_BANK_CONFLICT == 0
// measured (profiles/gemm8_pmc.md). Measured 1079 TF/s bf16 @4096^3 /
// 1031 @8192^3 (vs 652 TF for the 128^2 kernel; hipBLASLt 1645-2026;
// dense MFMA peak 2495). Two hard-won structural facts (the register
// cliff): the K-loop must be a counted SINGLE-EXIT loop (a mid-loop
// break kept extra live ranges across the exit merge: 146 spilled VGPRs,
// 178 TF), and per-site address math must reduce to scalar bases + two
// per-thread offsets.

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

namespace g8 {

typedef __bf16 bf16;
typedef bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

constexpr int BM = 256, BN = 256, BK = 64;
constexpr int HALF_ROWS = 128;                     // rows per half-tile
constexpr int HALF_ELEMS = HALF_ROWS * BK;         // 8192 bf16 = 16 KiB
// LDS slots: [dbuf(2)][operand A=0,B=1][half(2)] each HALF_ELEMS
__device__ __forceinline__ int lds_slot(int buf, int op, int half) {
  return ((buf * 2 + op) * 2 + half) * HALF_ELEMS;
}

// stage one half-tile: 512 threads x 2 chunks of 16 B, linear LDS dest,
// source pre-swizzled chunk = chunk ^ (row & 7). Addressing is decomposed
// into a SCALAR base (operand ptr + uniform row0*K + k0) plus ONE
// per-thread 64-bit offset per chunk (soff[i], computed once) — nothing
// per-site for the allocator to hoist and spill.
__device__ __forceinline__ void stage_half(bf16* lds_base, int slot, const bf16* base,
                                           const long* soff, int ldso) {
  const int t = threadIdx.x;
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const bf16* gptr = base + soff[i];
    bf16* dst = lds_base + slot + ldso + i * 512;
    __builtin_amdgcn_global_load_lds((const __attribute__((address_space(1))) unsigned int*)gptr,
                                     (__attribute__((address_space(3))) unsigned int*)dst, 16, 0, 0);
  }
  (void)t;
}

// swizzled ds_read of one A/B fragment: byte address = slot*2 + site
// constants (immediate) + row_byte (per-thread, precomputed) + (ck*16 ^
// lane7*16) — the XOR distributes over the *16 shift, so each site is one
// v_xor + one v_add on two per-thread registers.
__device__ __forceinline__ bf16x8 frag_read(const bf16* lds_base, int slot_byte, int row_byte,
                                            int ck16, int l7x16) {
  const int addr = slot_byte + row_byte + (ck16 ^ l7x16);
  return *reinterpret_cast<const bf16x8*>(reinterpret_cast<const char*>(lds_base) + addr);
}

#define BARRIER() __builtin_amdgcn_s_barrier()
#define LGKM0()                                   \
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory"); \
  __builtin_amdgcn_sched_barrier(0)
#define VMCNT4() asm volatile("s_waitcnt vmcnt(4)" ::: "memory")

__global__ __launch_bounds__(512, 2) void gemm8_bt_bf16_kernel(
    const bf16* __restrict__ A, const bf16* __restrict__ B,
    const bf16* __restrict__ bias, bf16* __restrict__ C, int M, int N, int K) {
  extern __shared__ bf16 lds[];
  const int nwg_n = N / BN;
  // T1: bijective XCD swizzle over the linear workgroup id
  const int nwg = gridDim.x;
  const int orig = blockIdx.x;
  const int q8 = nwg / 8, r8 = nwg % 8;
  const int xcd = orig % 8, pos = orig / 8;
  const int wgid = (xcd < r8 ? xcd * (q8 + 1) : r8 * (q8 + 1) + (xcd - r8) * q8) + pos;
  const int tile_m = wgid / nwg_n, tile_n = wgid % nwg_n;

  const int tid = threadIdx.x, lane = tid & 63, wid = tid >> 6;
  const int wr = wid >> 2, wc = wid & 3;  // 2M x 4N wave grid
  const int gm0 = tile_m * BM, gn0 = tile_n * BN;

  const int nkt = K / BK;  // number of K-tiles; K % 128 == 0 -> nkt even
  const int max_kt = nkt - 1;

  f32x4 acc[8][4];
#pragma unroll
  for (int mi = 0; mi < 8; ++mi)
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) acc[mi][ni] = f32x4{};

  // A rows for this wave live in half `wr`; B rows (n) in half `wc>>1`.
  // per-thread addressing state (the ONLY live address registers):
  const int a_row_byte = (lane & 15) * 128;                     // + mi*16*128 imm
  const int b_row_byte = ((wc & 1) * 64 + (lane & 15)) * 128;   // + ni*16*128 imm
  const int l7x16 = (lane & 7) * 16;
  const int g16 = (lane >> 4) * 16;                             // ck16 = kc*64 + g16
  long soff[2];
  const int ldso = (tid >> 6) * 1024;  // wave's LDS chunk base (elements)
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    const int q = (tid >> 6) * 128 + i * 64 + (tid & 63);  // physical chunk
    const int row = q >> 3;
    const int lc = (q & 7) ^ (row & 7);
    soff[i] = (long)row * K + lc * 8;
  }
  const bf16* Abase0 = A + (long)gm0 * K;
  const bf16* Abase1 = A + (long)(gm0 + 128) * K;
  const bf16* Bbase0 = B + (long)gn0 * K;
  const bf16* Bbase1 = B + (long)(gn0 + 128) * K;

  // ---- prologue: stage buf0.B, buf0.A (tile 0), buf1.B (tile 1) ----
  stage_half(lds, lds_slot(0, 1, 0), Bbase0, soff, ldso);
  stage_half(lds, lds_slot(0, 1, 1), Bbase1, soff, ldso);
  stage_half(lds, lds_slot(0, 0, 0), Abase0, soff, ldso);
  stage_half(lds, lds_slot(0, 0, 1), Abase1, soff, ldso);
  stage_half(lds, lds_slot(1, 1, 0), Bbase0 + BK, soff, ldso);
  stage_half(lds, lds_slot(1, 1, 1), Bbase1 + BK, soff, ldso);
  VMCNT4();
  BARRIER();

  bf16x8 afr[4], bfr[4][2];

  // one C-quadrant of 16 MFMA: m-halfrange mh (0:m0-3, 1:m4-7) x ONE kc —
  // afr holds a single (mh, kc) A-strip (re-read each phase, 4 reads);
  // bfr holds BOTH kc strips of all 4 n-tiles for the tile (persistent)
#define MFMA_QUAD(kc)                                                                  \
  _Pragma("unroll") for (int mi = 0; mi < 4; ++mi) {                                   \
    _Pragma("unroll") for (int ni = 0; ni < 4; ++ni) {                                 \
      acc[MH * 4 + mi][ni] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(                  \
          afr[mi], bfr[ni][kc], acc[MH * 4 + mi][ni], 0, 0, 0);                        \
    }                                                                                  \
  }

  const int a_slot_byte = lds_slot(0, 0, wr) * 2;      // + buf*4*HALF*2 imm
  const int b_slot_byte = lds_slot(0, 1, wc >> 1) * 2;
  constexpr int BUFB = 4 * HALF_ELEMS * 2;  // byte stride between dbuf slots

  // read this wave's A strip (4 m-tiles x one kc) from buf
#define READ_A(buf, mh, kc)                                                            \
  _Pragma("unroll") for (int mi = 0; mi < 4; ++mi) {                                   \
    afr[mi] = frag_read(lds, a_slot_byte + (buf) * BUFB + ((mh) * 64 + mi * 16) * 128, \
                        a_row_byte, (kc) * 64 + g16, l7x16);                           \
  }
  // read all 4 B n-tiles at one kc
#define READ_B(buf, kc)                                                                \
  _Pragma("unroll") for (int ni = 0; ni < 4; ++ni) {                                   \
    bfr[ni][kc] = frag_read(lds, b_slot_byte + (buf) * BUFB + ni * 16 * 128,           \
                            b_row_byte, (kc) * 64 + g16, l7x16);                       \
  }

// sched_barrier(0) pins each phase's code in place — without it the
// scheduler migrates reads/stages across phases and register pressure
// explodes past the 256-cap (HK technique list item 4)
#define PHASE(body_reads, stage_call, mfma_call, vm4)                                  \
  __builtin_amdgcn_sched_barrier(0);                                                   \
  body_reads;                                                                          \
  stage_call;                                                                          \
  BARRIER();                                                                           \
  LGKM0();                                                                             \
  __builtin_amdgcn_s_setprio(1);                                                       \
  mfma_call;                                                                           \
  __builtin_amdgcn_s_setprio(0);                                                       \
  __builtin_amdgcn_sched_barrier(0);                                                   \
  if (vm4) { VMCNT4(); }                                                               \
  BARRIER();

  // counted single-exit loop: K % 128 == 0 so nkt is even and every
  // iteration consumes exactly two K-tiles (multi-exit loops kept extra
  // live ranges across the exit merges)
  const int niter = nkt / 2;
  for (int it = 0; it < niter; ++it) {
    const int t2 = min(2 * it + 2, max_kt) * BK;  // stage k for tile 2it+2
    const int t3 = min(2 * it + 3, max_kt) * BK;  // tile 2it+3
    const int t1 = (2 * it + 1) * BK;             // tile 2it+1 (A staged now)

    // ---- phases 1-4: consume buf0 (tile 2it) ----
#define MH 0
    PHASE({ READ_A(0, 0, 0); READ_B(0, 0); },
          stage_half(lds, lds_slot(1, 0, 0), Abase0 + t1, soff, ldso), MFMA_QUAD(0), 0)
    PHASE({ READ_A(0, 0, 1); READ_B(0, 1); },
          stage_half(lds, lds_slot(1, 0, 1), Abase1 + t1, soff, ldso), MFMA_QUAD(1), 0)
#undef MH
#define MH 1
    PHASE({ READ_A(0, 1, 0); },
          stage_half(lds, lds_slot(0, 1, 0), Bbase0 + t2, soff, ldso), MFMA_QUAD(0), 0)
    PHASE({ READ_A(0, 1, 1); },
          stage_half(lds, lds_slot(0, 1, 1), Bbase1 + t2, soff, ldso), MFMA_QUAD(1), 1)
#undef MH

    // ---- phases 5-8: consume buf1 (tile 2it+1) ----
#define MH 0
    PHASE({ READ_A(1, 0, 0); READ_B(1, 0); },
          stage_half(lds, lds_slot(0, 0, 0), Abase0 + t2, soff, ldso), MFMA_QUAD(0), 0)
    PHASE({ READ_A(1, 0, 1); READ_B(1, 1); },
          stage_half(lds, lds_slot(0, 0, 1), Abase1 + t2, soff, ldso), MFMA_QUAD(1), 0)
#undef MH
#define MH 1
    PHASE({ READ_A(1, 1, 0); },
          stage_half(lds, lds_slot(1, 1, 0), Bbase0 + t3, soff, ldso), MFMA_QUAD(0), 0)
    PHASE({ READ_A(1, 1, 1); },
          stage_half(lds, lds_slot(1, 1, 1), Bbase1 + t3, soff, ldso), MFMA_QUAD(1), 1)
#undef MH
  }

  // ---- epilogue: C[m][n] bf16 ----
#pragma unroll
  for (int mi = 0; mi < 8; ++mi) {
#pragma unroll
    for (int ni = 0; ni < 4; ++ni) {
      const int n = gn0 + wc * 64 + ni * 16 + (lane & 15);
      const float bv = bias != nullptr ? (float)bias[n] : 0.f;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int m = gm0 + wr * 128 + mi * 16 + (lane >> 4) * 4 + j;
        C[(long)m * N + n] = (bf16)(acc[mi][ni][j] + bv);
      }
    }
  }
}


}  // namespace g8

// launcher used by bindings.hip
hipError_t launch_gemm8_bt(const void* A, const void* B, const void* bias, void* C,
                           int M, int N, int K, hipStream_t stream) {
  static bool attr_set = false;
  const int lds_bytes = 8 * g8::HALF_ELEMS * 2;  // 128 KiB
  if (!attr_set) {
    hipError_t e = hipFuncSetAttribute(reinterpret_cast<const void*>(&g8::gemm8_bt_bf16_kernel),
                                       hipFuncAttributeMaxDynamicSharedMemorySize, lds_bytes);
    if (e != hipSuccess) return e;
    attr_set = true;
  }
  dim3 grid((M / 256) * (N / 256));
  hipLaunchKernelGGL(g8::gemm8_bt_bf16_kernel, grid, dim3(512), lds_bytes, stream,
                     reinterpret_cast<const g8::bf16*>(A), reinterpret_cast<const g8::bf16*>(B),
                     reinterpret_cast<const g8::bf16*>(bias), reinterpret_cast<g8::bf16*>(C),
                     M, N, K);
  return hipGetLastError();
}
