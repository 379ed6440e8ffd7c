// Probe: discover the A/B fragment lane layouts of
// __builtin_amdgcn_mfma_f32_16x16x32_bf16 on gfx950 by testing layout
// hypotheses against a CPU reference (asymmetric random inputs — the CDNA4
// guide's G9 rule: symmetric tests cannot detect transposes).
//
// C/D layout is known/verified (guide §3): D[row][col], col = lane&15,
// row = (lane>>4)*4 + reg.
//
// Hypotheses for A (16 rows × 32 k) per lane l, elem j∈[0,8):
//   H0: row = l&15,  k = (l>>4)*8 + j          (contiguous k-chunk)
//   H1: row = l&15,  k = (l>>4) + 4*j           (strided k)
//   H2: row = l>>2?? (skipped — non-standard)
//   H3: row = l&15,  k = (l>>4)*4 + j + 16*(j>>2)  (two 4-blocks split 16 apart)
// B is hypothesized symmetric: col(N idx) = l&15, same k mapping.
// Build: hipcc --offload-arch=gfx950 -O2 mfma_layout_probe.hip -o probe
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>

typedef __bf16 bf16_t;
typedef bf16_t bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

__device__ __forceinline__ int k_of(int hyp, int l, int j) {
  switch (hyp) {
    case 0: return (l >> 4) * 8 + j;
    case 1: return (l >> 4) + 4 * j;
    case 3: return (l >> 4) * 4 + (j & 3) + 16 * (j >> 2);
    default: return j;
  }
}

template <int HYP>
__global__ void probe(const bf16_t* A, const bf16_t* Bt, float* D) {
  // A: [16][32] row-major ; Bt: [16][32] (N-major, k inner) ; D: [16][16]
  const int l = threadIdx.x;
  bf16x8 a{}, b{};
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = A[(l & 15) * 32 + k_of(HYP, l, j)];
    b[j] = Bt[(l & 15) * 32 + k_of(HYP, l, j)];
  }
  f32x4 c{};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    int row = (l >> 4) * 4 + j;
    int col = l & 15;
    D[row * 16 + col] = c[j];
  }
}

int main() {
  bf16_t *A, *Bt;
  float* D;
  hipMallocManaged(&A, 16 * 32 * sizeof(bf16_t));
  hipMallocManaged(&Bt, 16 * 32 * sizeof(bf16_t));
  hipMallocManaged(&D, 16 * 16 * sizeof(float));
  srand(7);
  float Af[16 * 32], Bf[16 * 32];
  for (int i = 0; i < 16 * 32; ++i) {
    Af[i] = (rand() % 17 - 8) * 0.25f;
    Bf[i] = (rand() % 13 - 6) * 0.5f;
    A[i] = (bf16_t)Af[i];
    Bt[i] = (bf16_t)Bf[i];
  }
  // CPU ref: C[m][n] = sum_k A[m][k] * Bt[n][k]
  float ref[16 * 16];
  for (int m = 0; m < 16; ++m)
    for (int n = 0; n < 16; ++n) {
      float s = 0;
      for (int k = 0; k < 32; ++k) s += Af[m * 32 + k] * Bf[n * 32 + k];
      ref[m * 16 + n] = s;
    }

  auto run = [&](int hyp, void (*kern)(const bf16_t*, const bf16_t*, float*)) {
    hipMemset(D, 0, 16 * 16 * sizeof(float));
    hipLaunchKernelGGL(kern, dim3(1), dim3(64), 0, 0, A, Bt, D);
    hipDeviceSynchronize();
    float maxd = 0;
    for (int i = 0; i < 256; ++i) maxd = fmaxf(maxd, fabsf(D[i] - ref[i]));
    // also check the transpose in case the C layout is flipped
    float maxdT = 0;
    for (int m = 0; m < 16; ++m)
      for (int n = 0; n < 16; ++n) maxdT = fmaxf(maxdT, fabsf(D[m * 16 + n] - ref[n * 16 + m]));
    printf("HYP %d: maxdiff=%f  maxdiff_vs_transpose=%f  %s\n", hyp, maxd, maxdT,
           maxd < 0.05 ? "<== MATCH" : (maxdT < 0.05 ? "<== TRANSPOSED MATCH" : ""));
  };
  run(0, probe<0>);
  run(1, probe<1>);
  run(3, probe<3>);
  return 0;
}
