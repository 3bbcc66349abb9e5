// Common device helpers for dlrover_amd CDNA4 (gfx950) kernels.
//
// Design per /opt/skills/guides/cdna_hip_programming.md:
//  - wave = 64 lanes (hard-coded, G: wavefront size);
//  - bf16 loads ALWAYS vectorized as short8/short4 reinterpret (Guideline 13);
//  - fp32 accumulation everywhere;
//  - memory-bound kernels use grid-stride loops capped at ~2048 blocks (G11).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>

#define WAVE_SIZE 64

typedef __attribute__((ext_vector_type(2))) short short2_t;
typedef __attribute__((ext_vector_type(4))) short short4_t;
typedef __attribute__((ext_vector_type(8))) short short8_t;
typedef __attribute__((ext_vector_type(4))) float float4_t;

using bf16_t = __hip_bfloat16;

// bf16 bits -> float (exact)
__device__ __forceinline__ float bf2f(short bits) {
  union { unsigned int u; float f; } cvt;
  cvt.u = ((unsigned int)(unsigned short)bits) << 16;
  return cvt.f;
}

// float -> bf16 bits, round-to-nearest-even
__device__ __forceinline__ short f2bf(float f) {
  union { unsigned int u; float f; } cvt;
  cvt.f = f;
  unsigned int u = cvt.u;
  unsigned int rounding = 0x7FFFu + ((u >> 16) & 1u);
  u += rounding;
  return (short)(u >> 16);
}

// Load 8 bf16 (16 B) and widen to fp32.
__device__ __forceinline__ void load8(const short* p, float out[8]) {
  short8_t v = *reinterpret_cast<const short8_t*>(p);
#pragma unroll
  for (int j = 0; j < 8; ++j) out[j] = bf2f(v[j]);
}

// Narrow 8 fp32 to bf16 and store as one 16 B write.
__device__ __forceinline__ void store8(short* p, const float in[8]) {
  short8_t v;
#pragma unroll
  for (int j = 0; j < 8; ++j) v[j] = f2bf(in[j]);
  *reinterpret_cast<short8_t*>(p) = v;
}

// ---------------------------------------------------------------------------
// Reductions. Wave-level via xor shuffle across all 64 lanes; block-level via
// LDS staging of per-wave partials (blockDim.x <= 1024 -> <= 16 waves).
// ---------------------------------------------------------------------------

__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE_SIZE);
  return v;
}

__device__ __forceinline__ float wave_reduce_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE_SIZE));
  return v;
}

// Block reduce; `scratch` must hold >= blockDim.x/64 floats. Result valid in
// ALL threads (broadcast via scratch).
__device__ __forceinline__ float block_reduce_sum(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x >> 6;
  const int nwaves = (blockDim.x + WAVE_SIZE - 1) >> 6;
  v = wave_reduce_sum(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float total = 0.f;
#pragma unroll 4
  for (int i = 0; i < nwaves; ++i) total += scratch[i];
  __syncthreads();
  return total;
}

__device__ __forceinline__ float block_reduce_max(float v, float* scratch) {
  const int lane = threadIdx.x & (WAVE_SIZE - 1);
  const int wid = threadIdx.x >> 6;
  const int nwaves = (blockDim.x + WAVE_SIZE - 1) >> 6;
  v = wave_reduce_max(v);
  if (lane == 0) scratch[wid] = v;
  __syncthreads();
  float m = -INFINITY;
#pragma unroll 4
  for (int i = 0; i < nwaves; ++i) m = fmaxf(m, scratch[i]);
  __syncthreads();
  return m;
}

// Grid sizing for memory-bound grid-stride kernels (Guideline 11):
// cap at 256 CU x 8 blocks = 2048 and stride the rest.
__host__ __forceinline__ int grid_capped(long long work_items, int block) {
  long long blocks = (work_items + block - 1) / block;
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

#define HIP_CHECK_KERNEL()                                                 \
  do {                                                                     \
    hipError_t e = hipGetLastError();                                      \
    if (e != hipSuccess) {                                                 \
      printf("dlrover_amd kernel launch failed: %s\n", hipGetErrorString(e)); \
    }                                                                      \
  } while (0)
