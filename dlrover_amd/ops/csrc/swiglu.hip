// Fused SwiGLU: out = silu(gate) * up, forward + backward, bf16 vectorized.
//
// Fusing the activation into one pass saves a full HBM round-trip of the
// intermediate (guide: fuse elementwise work into the producing kernel).
// gate/up are the two halves of the packed [N, 2*I] gate_up projection
// output (gate = [:, :I], up = [:, I:]) so the preceding GEMM stays a single
// hipBLASLt call.
#include "kern_common.h"

extern "C" {

__device__ __forceinline__ float silu(float x) {
  return x / (1.f + __expf(-x));
}

__global__ void swiglu_fwd_kernel(const short* __restrict__ gate_up,
                                  short* __restrict__ out,
                                  long long n_rows, int inner) {
  const int vecs = inner >> 3;
  const long long total = n_rows * vecs;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const long long row = i / vecs;
    const int v = (int)(i % vecs);
    const short* g = gate_up + row * (2LL * inner) + v * 8;
    const short* u = g + inner;
    float gv[8], uv[8], ov[8];
    load8(g, gv);
    load8(u, uv);
#pragma unroll
    for (int j = 0; j < 8; ++j) ov[j] = silu(gv[j]) * uv[j];
    store8(out + row * inner + v * 8, ov);
  }
}

// d_gate = dy * up * silu'(g);  silu'(g) = sig(g) * (1 + g * (1 - sig(g)))
// d_up   = dy * silu(g)
__global__ void swiglu_bwd_kernel(const short* __restrict__ dy,
                                  const short* __restrict__ gate_up,
                                  short* __restrict__ d_gate_up,
                                  long long n_rows, int inner) {
  const int vecs = inner >> 3;
  const long long total = n_rows * vecs;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < total; i += (long long)gridDim.x * blockDim.x) {
    const long long row = i / vecs;
    const int v = (int)(i % vecs);
    const short* g = gate_up + row * (2LL * inner) + v * 8;
    const short* u = g + inner;
    const short* dyp = dy + row * inner + v * 8;
    float gv[8], uv[8], dyv[8], dg[8], du[8];
    load8(g, gv);
    load8(u, uv);
    load8(dyp, dyv);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const float sig = 1.f / (1.f + __expf(-gv[j]));
      const float si = gv[j] * sig;
      dg[j] = dyv[j] * uv[j] * sig * (1.f + gv[j] * (1.f - sig));
      du[j] = dyv[j] * si;
    }
    short* dgp = d_gate_up + row * (2LL * inner) + v * 8;
    store8(dgp, dg);
    store8(dgp + inner, du);
  }
}

void swiglu_fwd_launch(const void* gate_up, void* out, long long n_rows,
                       int inner, hipStream_t stream) {
  const long long total = n_rows * (inner >> 3);
  hipLaunchKernelGGL(swiglu_fwd_kernel, dim3(grid_capped(total, 256)),
                     dim3(256), 0, stream, (const short*)gate_up, (short*)out,
                     n_rows, inner);
}

void swiglu_bwd_launch(const void* dy, const void* gate_up, void* d_gate_up,
                       long long n_rows, int inner, hipStream_t stream) {
  const long long total = n_rows * (inner >> 3);
  hipLaunchKernelGGL(swiglu_bwd_kernel, dim3(grid_capped(total, 256)),
                     dim3(256), 0, stream, (const short*)dy,
                     (const short*)gate_up, (short*)d_gate_up, n_rows, inner);
}

}  // extern "C"
