// Layout probe for the swapped-QK^T attention forward (CDNA4 8-warp ladder).
//
// Verifies, against plain-loop references computed on host, for ONE
// 32q x 64k x D=128 tile with random data:
//   B. swapped S^T = mfma(K, Q): per-lane value positions
//      sacc[kt][r] = S[q = lane&31][k = kt*32 + (r&3) + 8*(r>>2) + 4*(lane>>5)]
//   C. P relayout via bf16 pair-pack + permlane32_swap into PA fragments
//      pa[ks] elem jj = P[q = lane&31][k = ks*16 + (lane>>5)*8 + jj]
//   D. O = P·V via 4 mfma(A=pa[ks], B=V^T frag), C layout
//      col = d-tile*32 + (lane&31), row q = (r&3) + 8*(r>>2) + 4*(lane>>5)
//
// Build+run: hipcc --offload-arch=gfx950 -O2 fa_swapped_probe.hip -o p && ./p

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>

typedef __bf16 abf16;
typedef abf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef abf16 bf16x2 __attribute__((ext_vector_type(2)));
typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef unsigned uint2v __attribute__((ext_vector_type(2)));

__device__ __forceinline__ unsigned pack2(float lo, float hi) {
  union { bf16x2 h; unsigned u; } cv;
  cv.h = bf16x2{(abf16)lo, (abf16)hi};
  return cv.u;
}

// one wave: Q[32][128], K[64][128], V[64][128] row-major in global
__global__ void probe_kernel(const abf16* Q, const abf16* K, const abf16* V,
                             float* s_out /*[32][64]*/, float* p_out /*[4][64 slots]*/,
                             float* o_out /*[32][128]*/) {
  constexpr int D = 128;
  const int lane = threadIdx.x;
  const int hi = lane >> 5;
  const int col = lane & 31;

  // Q as B-operand fragments: lane holds Q row (lane&31), k-chunk t:
  // elements d = t*16 + hi*8 + jj
  bf16x8 qf[D / 16];
#pragma unroll
  for (int t = 0; t < D / 16; ++t)
    qf[t] = *reinterpret_cast<const bf16x8*>(Q + col * D + t * 16 + hi * 8);

  // K as A-operand fragments: kt picks 32 k-rows; lane holds K row
  // kt*32 + (lane&31), elements d = t*16 + hi*8 + jj
  f32x16 sacc[2] = {f32x16{}, f32x16{}};
#pragma unroll
  for (int kt = 0; kt < 2; ++kt)
#pragma unroll
    for (int t = 0; t < D / 16; ++t) {
      bf16x8 kf = *reinterpret_cast<const bf16x8*>(K + (kt * 32 + col) * D + t * 16 + hi * 8);
      sacc[kt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[t], sacc[kt], 0, 0, 0);
    }

  // stage B dump: S[q][k]
#pragma unroll
  for (int kt = 0; kt < 2; ++kt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int k = kt * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
      s_out[col * 64 + k] = sacc[kt][r];
    }

  // stage C: relayout P (use raw S values, bf16-rounded) into PA fragments
  unsigned pa[4][4];  // [ks][word] — words jj=(0,1),(2,3),(4,5),(6,7)
#pragma unroll
  for (int ks = 0; ks < 4; ++ks) {
    const f32x16& p = sacc[ks >> 1];
    const int base = 8 * (ks & 1);
    uint2v r01 = __builtin_amdgcn_permlane32_swap(
        pack2(p[base + 0], p[base + 1]), pack2(p[base + 4], p[base + 5]), false, false);
    uint2v r23 = __builtin_amdgcn_permlane32_swap(
        pack2(p[base + 2], p[base + 3]), pack2(p[base + 6], p[base + 7]), false, false);
    pa[ks][0] = r01.x;  // jj = 0,1
    pa[ks][1] = r23.x;  // jj = 2,3
    pa[ks][2] = r01.y;  // jj = 4,5
    pa[ks][3] = r23.y;  // jj = 6,7
  }
  // dump PA fragments as floats: p_out[ks][lane*? ] layout: for each ks,
  // lane, jj: P[col][ks*16 + hi*8 + jj] — write to [32][64] grid to compare
#pragma unroll
  for (int ks = 0; ks < 4; ++ks)
#pragma unroll
    for (int w = 0; w < 4; ++w) {
      union { unsigned u; bf16x2 h; } cv;
      cv.u = pa[ks][w];
      int k0 = ks * 16 + hi * 8 + 2 * w;
      p_out[col * 64 + k0] = (float)cv.h[0];
      p_out[col * 64 + k0 + 1] = (float)cv.h[1];
    }

  // stage D: O = P·V via mfma(A=pa[ks], B = V^T rows d)
  f32x16 oacc[D / 32] = {};
#pragma unroll
  for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      // B-frag: lane holds V^T row d = dt*32 + col, elements
      // k = ks*16 + hi*8 + jj  -> V[k][d] for 8 consecutive k
      abf16 vt[8];
#pragma unroll
      for (int jj = 0; jj < 8; ++jj)
        vt[jj] = V[(ks * 16 + hi * 8 + jj) * D + dt * 32 + col];
      bf16x8 vf = *reinterpret_cast<bf16x8*>(vt);
      oacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
          *reinterpret_cast<bf16x8*>(pa[ks]), vf, oacc[dt], 0, 0, 0);
    }
#pragma unroll
  for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      int q = (r & 3) + 8 * (r >> 2) + 4 * hi;
      o_out[q * D + dt * 32 + col] = oacc[dt][r];
    }
}

// ---------------------------------------------------------------------------
// Stage E (separate kernel): ds_read_b64_tr_b16 B-fragment path for PV.
// V stored in LDS as [k/4][d/16][4][16] subtiles; each 16-lane group's
// addresses tile one 128 B subtile window (lane slot = (l&15)*8 B) and the
// transpose delivers column (l&15): lane l gets
// V[k0 + 8*(l>>5) + j][d0 + (l&31)] for j=0..3 (second read +512 elements
// covers j=4..7). Verified against the direct loads.
// ---------------------------------------------------------------------------

typedef __bf16 bf16x4p __attribute__((ext_vector_type(4)));

__global__ void probe_tr_kernel(const abf16* V, float* err_count) {
  constexpr int D = 128;
  __shared__ abf16 Vt[64 * 128];
  const int lane = threadIdx.x;
  const int hi = lane >> 5;
  const int col = lane & 31;
  // stage V[64][128] into subtiled layout
  for (int i = lane; i < 64 * (D / 8); i += 64) {
    const int k = i / (D / 8);
    const int d0 = (i % (D / 8)) * 8;
    const int base = ((k >> 2) * 8 + (d0 >> 4)) * 64 + (k & 3) * 16 + (d0 & 15);
    *reinterpret_cast<bf16x8*>(Vt + base) = *reinterpret_cast<const bf16x8*>(V + k * D + d0);
  }
  __syncthreads();
  int bad = 0;
  for (int dt = 0; dt < D / 32; ++dt)
    for (int ks = 0; ks < 4; ++ks) {
      const int k0 = ks * 16;
      const int d0 = dt * 32;
      // per-lane slot: 16-lane groups tile a 128 B window; lane l reads
      // column (l&15) of subtile (kb = k0/4 + 2*hi, db = d0/16 + ((l>>4)&1))
      const unsigned addr_elems =
          64u * ((k0 / 4 + 2 * hi) * 8 + d0 / 16 + ((lane >> 4) & 1)) + (lane & 15) * 4;
      auto p0 = (__attribute__((address_space(3))) bf16x4p*)(Vt + addr_elems);
      auto p1 = (__attribute__((address_space(3))) bf16x4p*)(Vt + addr_elems + 512);
      bf16x4p lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p0);
      bf16x4p hi4 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(p1);
      for (int j = 0; j < 4; ++j) {
        float want_lo = (float)V[(k0 + hi * 8 + j) * D + d0 + col];
        float want_hi = (float)V[(k0 + 4 + hi * 8 + j) * D + d0 + col];
        if ((float)lo[j] != want_lo) ++bad;
        if ((float)hi4[j] != want_hi) ++bad;
      }
    }
  atomicAdd(err_count, (float)bad);
}

int main() {
  constexpr int D = 128;
  srand(7);
  auto frand = []() { return (float)(rand() % 2000 - 1000) / 500.f; };
  abf16 *Q, *K, *V;
  float *S, *P, *O;
  hipMallocManaged(&Q, 32 * D * sizeof(abf16));
  hipMallocManaged(&K, 64 * D * sizeof(abf16));
  hipMallocManaged(&V, 64 * D * sizeof(abf16));
  hipMallocManaged(&S, 32 * 64 * sizeof(float));
  hipMallocManaged(&P, 32 * 64 * sizeof(float));
  hipMallocManaged(&O, 32 * D * sizeof(float));
  for (int i = 0; i < 32 * D; ++i) Q[i] = (abf16)frand();
  for (int i = 0; i < 64 * D; ++i) K[i] = (abf16)frand();
  for (int i = 0; i < 64 * D; ++i) V[i] = (abf16)frand();

  hipLaunchKernelGGL(probe_kernel, dim3(1), dim3(64), 0, 0, Q, K, V, S, P, O);
  float* errc;
  hipMallocManaged(&errc, sizeof(float));
  *errc = 0.f;
  hipLaunchKernelGGL(probe_tr_kernel, dim3(1), dim3(64), 0, 0, V, errc);
  hipError_t err = hipDeviceSynchronize();
  if (err != hipSuccess) { printf("HIP ERROR: %s\n", hipGetErrorString(err)); return 2; }
  printf("stage E (tr_read V subtile): %s (%g bad)\n", *errc == 0.f ? "PASS" : "FAIL", *errc);

  // host references
  int bad = 0;
  float sref[32][64];
  for (int q = 0; q < 32; ++q)
    for (int k = 0; k < 64; ++k) {
      float acc = 0;
      for (int d = 0; d < D; ++d) acc += (float)Q[q * D + d] * (float)K[k * D + d];
      sref[q][k] = acc;
      if (fabsf(acc - S[q * 64 + k]) > 0.02f * fmaxf(1.f, fabsf(acc))) {
        if (bad++ < 5) printf("S mismatch q%d k%d: got %f want %f\n", q, k, S[q * 64 + k], acc);
      }
    }
  printf("stage B (swapped S^T): %s\n", bad ? "FAIL" : "PASS");

  int badp = 0;
  for (int q = 0; q < 32; ++q)
    for (int k = 0; k < 64; ++k) {
      float want = (float)(abf16)sref[q][k];  // bf16-rounded
      if (fabsf(want - P[q * 64 + k]) > 0.02f * fmaxf(1.f, fabsf(want))) {
        if (badp++ < 5) printf("P mismatch q%d k%d: got %f want %f\n", q, k, P[q * 64 + k], want);
      }
    }
  printf("stage C (pack+permlane relayout): %s\n", badp ? "FAIL" : "PASS");

  int bado = 0;
  for (int q = 0; q < 32; ++q)
    for (int d = 0; d < D; ++d) {
      float acc = 0;
      for (int k = 0; k < 64; ++k) acc += (float)(abf16)sref[q][k] * (float)V[k * D + d];
      if (fabsf(acc - O[q * D + d]) > 0.05f * fmaxf(1.f, fabsf(acc))) {
        if (bado++ < 5) printf("O mismatch q%d d%d: got %f want %f\n", q, d, O[q * D + d], acc);
      }
    }
  printf("stage D (PV via PA frags): %s\n", bado ? "FAIL" : "PASS");
  return (bad || badp || bado) ? 1 : 0;
}

