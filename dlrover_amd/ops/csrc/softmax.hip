// Fused scale + causal-mask + softmax over attention scores, in-place, bf16.
//
// v1 attention path: scores = Q @ K^T via hipBLASLt (library GEMM), then THIS
// kernel does scale+mask+softmax in one HBM pass instead of three eager ops
// (mask add, softmax, dtype casts). Backward computes
// dS = P * (dP - sum(dP * P)) fused the same way.
//
// scores layout: [n_rows, row_len] where row r belongs to query position
// q = (r % q_len) + q_offset; causal-valid columns are [0, q+1).
#include "kern_common.h"

extern "C" {

__global__ void causal_softmax_fwd_kernel(
    short* __restrict__ scores, long long n_rows, int row_len, int q_len,
    int q_offset, float scale) {
  __shared__ float scratch[16];
  const int vecs = row_len >> 3;
  for (long long row = blockIdx.x; row < n_rows; row += gridDim.x) {
    short* r = scores + row * row_len;
    const int qpos = (int)(row % q_len) + q_offset;
    const int valid = qpos + 1 < row_len ? qpos + 1 : row_len;
    // pass 1: max over valid prefix
    float mx = -INFINITY;
    for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
      if (v * 8 >= valid) break;
      float xv[8];
      load8(r + v * 8, xv);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (v * 8 + j < valid) mx = fmaxf(mx, xv[j] * scale);
    }
    mx = block_reduce_max(mx, scratch);
    // pass 2: sumexp
    float sum = 0.f;
    for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
      if (v * 8 >= valid) break;
      float xv[8];
      load8(r + v * 8, xv);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        if (v * 8 + j < valid) sum += __expf(xv[j] * scale - mx);
    }
    sum = block_reduce_sum(sum, scratch);
    const float inv = 1.f / sum;
    // pass 3: write normalized probs (zeros beyond the causal boundary)
    for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
      float xv[8], ov[8];
      load8(r + v * 8, xv);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        ov[j] = (v * 8 + j < valid) ? __expf(xv[j] * scale - mx) * inv : 0.f;
      store8(r + v * 8, ov);
    }
    __syncthreads();
  }
}

// dS = P * (dP - rowdot) * scale, where rowdot = sum_j dP_j * P_j.
// dP arrives in `dscores` and is overwritten with dS.
__global__ void causal_softmax_bwd_kernel(
    short* __restrict__ dscores, const short* __restrict__ probs,
    long long n_rows, int row_len, float scale) {
  __shared__ float scratch[16];
  const int vecs = row_len >> 3;
  for (long long row = blockIdx.x; row < n_rows; row += gridDim.x) {
    short* dr = dscores + row * row_len;
    const short* pr = probs + row * row_len;
    float dot = 0.f;
    for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
      float dv[8], pv[8];
      load8(dr + v * 8, dv);
      load8(pr + v * 8, pv);
#pragma unroll
      for (int j = 0; j < 8; ++j) dot += dv[j] * pv[j];
    }
    dot = block_reduce_sum(dot, scratch);
    for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
      float dv[8], pv[8], ov[8];
      load8(dr + v * 8, dv);
      load8(pr + v * 8, pv);
#pragma unroll
      for (int j = 0; j < 8; ++j) ov[j] = pv[j] * (dv[j] - dot) * scale;
      store8(dr + v * 8, ov);
    }
    __syncthreads();
  }
}

void causal_softmax_fwd_launch(void* scores, long long n_rows, int row_len,
                               int q_len, int q_offset, float scale,
                               hipStream_t stream) {
  int grid = n_rows < 2048 ? (int)n_rows : 2048;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(causal_softmax_fwd_kernel, dim3(grid), dim3(256), 0,
                     stream, (short*)scores, n_rows, row_len, q_len, q_offset,
                     scale);
}

void causal_softmax_bwd_launch(void* dscores, const void* probs,
                               long long n_rows, int row_len, float scale,
                               hipStream_t stream) {
  int grid = n_rows < 2048 ? (int)n_rows : 2048;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(causal_softmax_bwd_kernel, dim3(grid), dim3(256), 0,
                     stream, (short*)dscores, (const short*)probs, n_rows,
                     row_len, scale);
}

}  // extern "C"
