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

// ---- 32x32x16 MFMA layout probe ---------------------------------------------
//
// C[32,32] = A[32,16] @ B[16,32] with one v_mfma_f32_32x32x16_bf16 per wave.
// Assumed layouts (guide §3 gives C/D measured; A/B by analogy with the
// 16x16x32 family — THIS PROBE verifies them on hardware before the FA v3
// kernel builds on them):
//   A[32,16]: lane l holds A[l%32][(l/32)*8 + i], i in [0,8)
//   B[16,32]: lane l holds B[(l/32)*8 + i][l%32]
//   C[32,32]: lane l reg r -> C[(r&3) + 8*(r>>2) + 4*(l>>5)][l&31]
//             (guide §3, measured m74/m101), r in [0,16)
__global__ void mfma32_probe_kernel(const short* __restrict__ A,
                                    const short* __restrict__ B,
                                    float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  const int half = lane >> 5;  // which 8-wide k-slice
  const int sub = lane & 31;
  bf16x8 a, b;
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    short abits = A[sub * 16 + half * 8 + i];
    short bbits = B[(half * 8 + i) * 32 + sub];
    a[i] = *reinterpret_cast<__bf16*>(&abits);
    b[i] = *reinterpret_cast<__bf16*>(&bbits);
  }
  typedef __attribute__((ext_vector_type(16))) float f32x16_;
  f32x16_ acc;
#pragma unroll
  for (int r = 0; r < 16; ++r) acc[r] = 0.f;
  acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
      *reinterpret_cast<bf16x8*>(&a), *reinterpret_cast<bf16x8*>(&b), acc, 0,
      0, 0);
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    const int row = (r & 3) + 8 * (r >> 2) + 4 * half;
    C[row * 32 + sub] = acc[r];
  }
}

extern "C" void mfma32_probe_launch(const void* A, const void* B, void* C,
                                    hipStream_t stream) {
  hipLaunchKernelGGL(mfma32_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const short*)A, (const short*)B, (float*)C);
}

// ---- ds_read_b64_tr_b16 semantics probe -------------------------------------
//
// Hardware-measured semantics (two probe rounds on MI355X):
//   * Per-lane address bits 1-2 select the COLUMN (A/2 & 3) of a row-major
//     4x4 bf16 tile; the lane receives that column (4 elements, +4 apart).
//   * The tile itself is NOT selected per-lane: using per-lane high bits in
//     the FA v2 kernel read the group leader's tile for every lane
//     (cols with sub>3 got sub&3's data). This probe varies a high address
//     bit WITHIN a group (odd sub lanes +32B) to pin down which bits are
//     taken per-lane vs from the wave/group.
//   * offset:N is additive.
// Round-2 FA work must build on whatever `expect_per_lane` vs
// `expect_leader` this probe reports (tests/test_mfma_gpu.py).
__global__ void tr_b16_probe_kernel(short* __restrict__ out) {
  __shared__ short lds[1024];
  const int lane = threadIdx.x & 63;
  for (int i = threadIdx.x; i < 1024; i += blockDim.x) {
    float v = (float)i;
    lds[i] = f2bf(v);
  }
  __syncthreads();
  typedef __attribute__((ext_vector_type(2))) unsigned int u32x2;
  u32x2 r0, r1, r2;
  // addr pattern A: the original uniform-high-bits layout (2*l)
  const unsigned addrA = (unsigned)(uintptr_t)&lds[0] + 2 * lane;
  // addr pattern B: odd-sub lanes point one 4x4 tile (32 B) further —
  // distinguishes per-lane high bits from group-leader high bits
  const unsigned addrB = addrA + ((lane & 1) ? 32u : 0u);
  asm volatile("ds_read_b64_tr_b16 %0, %3 offset:0\n\t"
               "ds_read_b64_tr_b16 %1, %3 offset:128\n\t"
               "ds_read_b64_tr_b16 %2, %4 offset:0\n\t"
               "s_waitcnt lgkmcnt(0)"
               : "=v"(r0), "=v"(r1), "=v"(r2)
               : "v"(addrA), "v"(addrB));
  short vals[12];
  *reinterpret_cast<u32x2*>(&vals[0]) = r0;
  *reinterpret_cast<u32x2*>(&vals[4]) = r1;
  *reinterpret_cast<u32x2*>(&vals[8]) = r2;
#pragma unroll
  for (int j = 0; j < 12; ++j) out[lane * 12 + j] = vals[j];
}

extern "C" void tr_b16_probe_launch(void* out, hipStream_t stream) {
  hipLaunchKernelGGL(tr_b16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (short*)out);
}

// ---- tr_b16 probe r3: COOPERATIVE tile sourcing ----------------------------
//
// Round-1 established: per-lane bits 1-2 select the column of a 4x4 tile
// (rows at +0,+8,+16,+24 bytes), tile-high bits are NOT taken per-lane, and
// with mixed high bits the HW sources them "cooperatively". AITER/HK issue
// tr reads with per-lane-VARYING tile addresses productively (192 reads per
// loop, not 4x that), so the cooperative mode must be structured. This probe
// maps it exactly: LDS word i holds raw bits i, so each returned value
// identifies (tile, row, col) it came from.
//   read A: tile base varies per LANE   (addr = lane*32)       — full map
//   read B: tile base varies per QUAD   (addr = (lane>>2)*32 + (lane&3)*2)
//   read C: base uniform per 16-group, column per lane, offset immediate
__global__ void tr_b16_probe3_kernel(short* __restrict__ out) {
  __shared__ short lds[2048];
  for (int i = threadIdx.x; i < 2048; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  const int lane = threadIdx.x & 63;
  typedef __attribute__((ext_vector_type(2))) unsigned int u32x2;
  const unsigned base = (unsigned)(uintptr_t)&lds[0];
  const unsigned addrA = base + (unsigned)lane * 32u;
  const unsigned addrB = base + (unsigned)(lane >> 2) * 32u + (lane & 3) * 2u;
  const unsigned addrC = base + (unsigned)(lane >> 4) * 512u + (lane & 3) * 2u;
  u32x2 rA, rB, rC;
  asm volatile("ds_read_b64_tr_b16 %0, %3\n\t"
               "ds_read_b64_tr_b16 %1, %4\n\t"
               "ds_read_b64_tr_b16 %2, %5 offset:64\n\t"
               "s_waitcnt lgkmcnt(0)"
               : "=v"(rA), "=v"(rB), "=v"(rC)
               : "v"(addrA), "v"(addrB), "v"(addrC));
  short vals[12];
  *reinterpret_cast<u32x2*>(&vals[0]) = rA;
  *reinterpret_cast<u32x2*>(&vals[4]) = rB;
  *reinterpret_cast<u32x2*>(&vals[8]) = rC;
#pragma unroll
  for (int j = 0; j < 12; ++j) out[lane * 12 + j] = vals[j];
}

extern "C" void tr_b16_probe3_launch(void* out, hipStream_t stream) {
  hipLaunchKernelGGL(tr_b16_probe3_kernel, dim3(1), dim3(64), 0, stream,
                     (short*)out);
}
