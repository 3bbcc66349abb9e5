// MFMA layout self-test: C[16,16] = A[16,32] @ B[32,16] with one
// v_mfma_f32_16x16x32_bf16 per wave.
//
// Assumed gfx950 operand layouts (to be verified on hardware against a torch
// reference with random ASYMMETRIC inputs — guide §3 ERRATA: symmetric tests
// can't catch transposes):
//   A[16,32]: lane l holds A[l%16][(l/16)*8 + i], i in [0,8)   (8 bf16)
//   B[32,16]: lane l holds B[(l/16)*8 + i][l%16]
//   C[16,16]: lane l reg r -> C[(l>>4)*4 + r][l&15]  (verified, guide §3)
// The flash-attention kernels build on exactly these mappings.
#include "kern_common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4_;

__global__ void mfma16_probe_kernel(const short* __restrict__ A,
                                    const short* __restrict__ B,
                                    float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  const int half = lane >> 4;  // which 8-wide k-slice this lane holds
  const int sub = lane & 15;
  bf16x8 a, b;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    short abits = A[(sub)*32 + half * 8 + i];
    short bbits = B[(half * 8 + i) * 16 + sub];
    a[i] = *reinterpret_cast<__bf16*>(&abits);
    b[i] = *reinterpret_cast<__bf16*>(&bbits);
  }
  f32x4_ acc = {0.f, 0.f, 0.f, 0.f};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
      *reinterpret_cast<bf16x8*>(&a), *reinterpret_cast<bf16x8*>(&b), acc, 0,
      0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) C[((lane >> 4) * 4 + r) * 16 + (lane & 15)] = acc[r];
}

extern "C" void mfma16_probe_launch(const void* A, const void* B, void* C,
                                    hipStream_t stream) {
  hipLaunchKernelGGL(mfma16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const short*)A, (const short*)B, (float*)C);
}
