// MX-fp8 MFMA layout probe for gfx950: verify the operand/scale layout of
// __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4 against a CPU reference.
//
// Hypotheses under test (extending the verified 32x32x16_bf16 mapping):
//   A: lane holds row = lane&31, k = (lane>>5)*32 + j  (32 contiguous fp8)
//   B: lane holds col = lane&31, k = (lane>>5)*32 + j
//   C/D: col = lane&31, row = (reg&3) + 8*(reg>>2) + 4*(lane>>5)
//   scale: ONE e8m0 byte per lane (its k-block), low byte of the int arg
//          (127 = 2^0); opsel args select the byte.
// Asymmetric random inputs (transpose-detecting, guide G9).

#include <hip/hip_runtime.h>
#include <hip/hip_fp8.h>
#include <cstdio>
#include <cstdlib>
#include <cmath>
#include <vector>

typedef float f32x16 __attribute__((ext_vector_type(16)));
typedef int i32x8 __attribute__((ext_vector_type(8)));

__global__ void mx_probe(const unsigned char* A, const unsigned char* B, float* D,
                         int scale_a_byte, int scale_b_byte) {
  const int lane = threadIdx.x & 63;
  // scale-mapping decode modes: scale_a_byte < 0 selects a pattern
  //  -1: sa = 127 + (lane & 7)   (row-dependent)
  //  -2: sa = 127 + (lane >> 5)  (k-block-dependent)
  int sa_eff = scale_a_byte, sb_eff = scale_b_byte;
  if (scale_a_byte == -1) sa_eff = 127 + (lane & 7);
  if (scale_a_byte == -2) sa_eff = 127 + (lane >> 5);
  if (scale_a_byte == -3) sa_eff = 127 + ((lane >> 3) & 3);
  if (scale_a_byte == -4) sa_eff = 127 + ((lane >> 5) & 1) * 3;
  i32x8 a, b;
  const int g = lane >> 5;
#pragma unroll
  for (int r = 0; r < 8; ++r) {
    int av = 0, bv = 0;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int k = g * 32 + r * 4 + c;
      av |= (int)A[(lane & 31) * 64 + k] << (8 * c);
      bv |= (int)B[(lane & 31) * 64 + k] << (8 * c);
    }
    a[r] = av;
    b[r] = bv;
  }
  f32x16 acc = {};
  // fmt 0 = fp8 e4m3 for both operands; scales in byte 0 (opsel 0,0)
  if (scale_a_byte == -5) {
    // chaining semantics: acc1 = mfma(1,1,0,s=1.0) -> 32; then
    // acc2 = mfma(1,1,acc1,s=2.0): CORRECT (scale only A*B) -> 32+64=96;
    // BROKEN (scale applied to sum) -> (32+32)*2 = 128
    acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, acc, 0, 0, 0, 127, 0, 127);
    acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, acc, 0, 0, 0, 128, 0, 127);
  } else if (scale_a_byte == -6) {
    // chained with DEPENDENCY-PINNED nops between (plain asm volatile gets
    // hoisted above both calls)
    acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, acc, 0, 0, 0, 127, 0, 127);
    asm volatile("s_nop 15" : "+v"(acc));
    acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, acc, 0, 0, 0, 128, 0, 127);
  } else if (scale_a_byte == -7) {
    // separate accumulators: report acc in D rows, acc2 in D+1024
    f32x16 acc2 = {};
    acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, acc, 0, 0, 0, 127, 0, 127);
    acc2 = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, acc2, 0, 0, 0, 128, 0, 127);
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int row = (r & 3) + 8 * (r >> 2) + 4 * g;
      D[1024 + row * 32 + (lane & 31)] = acc2[r];
    }
  } else if (scale_a_byte == -8) {
    // THREE chained, scales 1x,2x,4x: effective per-call scales reveal the
    // ld_scale->mfma binding shift. correct => 32+64+128 = 224
    acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, acc, 0, 0, 0, 127, 0, 127);
    acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, acc, 0, 0, 0, 128, 0, 127);
    acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, acc, 0, 0, 0, 129, 0, 127);
  } else {
    acc = __builtin_amdgcn_mfma_scale_f32_32x32x64_f8f6f4(a, b, acc, 0, 0, 0, sa_eff, 0,
                                                          sb_eff);
  }
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * g;
    D[row * 32 + (lane & 31)] = acc[r];
  }
}

int main() {
  std::vector<unsigned char> hA(32 * 64), hB(32 * 64);
  std::vector<float> fA(32 * 64), fB(32 * 64);
  srand(3);
  for (int i = 0; i < 32 * 64; ++i) {
    float v = (rand() % 17 - 8) * 0.25f;
    __hip_fp8_e4m3 q(v);
    hA[i] = *reinterpret_cast<unsigned char*>(&q);
    fA[i] = (float)q;
    v = (rand() % 17 - 8) * 0.25f;
    __hip_fp8_e4m3 q2(v);
    hB[i] = *reinterpret_cast<unsigned char*>(&q2);
    fB[i] = (float)q2;
  }
  unsigned char *dA, *dB;
  float* dD;
  hipMalloc(&dA, 32 * 64);
  hipMalloc(&dB, 32 * 64);
  hipMalloc(&dD, 2 * 32 * 32 * 4);
  hipMemcpy(dA, hA.data(), 32 * 64, hipMemcpyHostToDevice);
  hipMemcpy(dB, hB.data(), 32 * 64, hipMemcpyHostToDevice);

  auto run = [&](int sa, int sb, const char* tag, float expect_mult) {
    hipLaunchKernelGGL(mx_probe, dim3(1), dim3(64), 0, 0, dA, dB, dD, sa, sb);
    hipDeviceSynchronize();
    std::vector<float> hD(32 * 32);
    hipMemcpy(hD.data(), dD, 32 * 32 * 4, hipMemcpyDeviceToHost);
    double worst = 0;
    int bad = 0;
    for (int m = 0; m < 32; ++m)
      for (int n = 0; n < 32; ++n) {
        float ref = 0;
        for (int k = 0; k < 64; ++k) ref += fA[m * 64 + k] * fB[n * 64 + k];
        ref *= expect_mult;
        float got = hD[m * 32 + n];
        double rel = fabs(got - ref) / (fabs(ref) + 1e-3);
        if (rel > worst) worst = rel;
        if (rel > 0.05) ++bad;
      }
    printf("%s: worst rel %.4f, bad %d/1024  (D[0][0]=%f D[5][7]=%f)\n", tag, worst, bad,
           hD[0], hD[5 * 32 + 7]);
  };

  run(127, 127, "scale=1.0 both (byte0=127)", 1.0f);
  run(128, 127, "scale_a=2.0 (byte0=128)", 2.0f);
  run(127, 129, "scale_b=4.0 (byte0=129)", 4.0f);

  // mapping decode: ones data, patterned scales; dump row 0/1/8 cols 0..3
  std::vector<unsigned char> ones(32 * 64);
  __hip_fp8_e4m3 one(1.0f);
  for (auto& u : ones) u = *reinterpret_cast<unsigned char*>(&one);
  hipMemcpy(dA, ones.data(), 32 * 64, hipMemcpyHostToDevice);
  hipMemcpy(dB, ones.data(), 32 * 64, hipMemcpyHostToDevice);
  auto dump = [&](int sa, const char* tag) {
    hipLaunchKernelGGL(mx_probe, dim3(1), dim3(64), 0, 0, dA, dB, dD, sa, 127);
    hipDeviceSynchronize();
    std::vector<float> hD(2 * 32 * 32);
    hipMemcpy(hD.data(), dD, 2 * 32 * 32 * 4, hipMemcpyDeviceToHost);
    printf("%s:\n", tag);
    for (int m : {0, 1, 2, 7, 8, 9}) printf("  D[%d][0]=%g D[%d][17]=%g\n", m, hD[m * 32], m, hD[m * 32 + 17]);
    if (sa == -7) printf("  acc2[0][0]=%g acc2[5][7]=%g\n", hD[1024], hD[1024 + 5 * 32 + 7]);
  };
  dump(-1, "sa=127+(lane&7) [row-dep?]   expect D[m][*]=64*2^(m&7) if row-mapped");
  dump(-2, "sa=127+(lane>>5) [g-dep?]    expect D[*][*]=32*(1+2)=96 if g-mapped");
  dump(-3, "sa=127+((lane>>3)&3)         expect D[m][*]=64*2^((m>>3)&3) if lane==row");
  dump(-5, "chained acc, s1=1 s2=2       96 = correct (scale A*B only); 128 = scale hits acc");
  dump(-6, "chained + s_nop separation   expect 96");
  dump(-7, "separate accs + VALU add     expect 96");
  dump(-8, "3 chained s=1,2,4            correct=224; shift-by-one(2,4,4)=352; (2,4,1)=256");
  dump(-4, "sa=127+((lane>>5)&1)*3       expect D[*][*]=32*(1+8)=288 if g=lane>>5");
  return 0;
}
