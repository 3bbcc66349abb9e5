// Fused cross-entropy: bf16 logits [N, V] -> per-token fp32 loss AND dlogits
// written back in place, in one kernel.
//
// At Llama-3 vocab (128256) the logits tensor dominates activation memory;
// eager CE does softmax + gather + backward as separate HBM passes over
// ~1 GB. This kernel does max / sumexp / (P - onehot) in three strided passes
// through L2 per row and never materializes a separate probability tensor.
// ignore_index rows produce loss 0 and zero grads.
#include "kern_common.h"

extern "C" {

__global__ void cross_entropy_fwd_bwd_kernel(
    short* __restrict__ logits, const int* __restrict__ targets,
    float* __restrict__ losses, long long n_rows, int vocab, int ignore_index,
    float grad_scale, int compute_grad) {
  __shared__ float scratch[16];
  const int vecs = vocab >> 3;
  const int tail0 = vecs * 8;
  for (long long row = blockIdx.x; row < n_rows; row += gridDim.x) {
    short* r = logits + row * vocab;
    const int tgt = targets[row];
    if (tgt == ignore_index) {
      if (compute_grad) {
        for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
          float z[8] = {0, 0, 0, 0, 0, 0, 0, 0};
          store8(r + v * 8, z);
        }
        for (int j = tail0 + threadIdx.x; j < vocab; j += blockDim.x)
          r[j] = f2bf(0.f);
      }
      if (threadIdx.x == 0) losses[row] = 0.f;
      __syncthreads();
      continue;
    }
    float mx = -INFINITY;
    for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
      float xv[8];
      load8(r + v * 8, xv);
#pragma unroll
      for (int j = 0; j < 8; ++j) mx = fmaxf(mx, xv[j]);
    }
    for (int j = tail0 + threadIdx.x; j < vocab; j += blockDim.x)
      mx = fmaxf(mx, bf2f(r[j]));
    mx = block_reduce_max(mx, scratch);
    float sum = 0.f;
    for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
      float xv[8];
      load8(r + v * 8, xv);
#pragma unroll
      for (int j = 0; j < 8; ++j) sum += __expf(xv[j] - mx);
    }
    for (int j = tail0 + threadIdx.x; j < vocab; j += blockDim.x)
      sum += __expf(bf2f(r[j]) - mx);
    sum = block_reduce_sum(sum, scratch);
    if (threadIdx.x == 0)
      losses[row] = logf(sum) + mx - bf2f(r[tgt]);
    // the grad pass overwrites r; the loss read of r[tgt] must land first
    __syncthreads();
    if (compute_grad) {
      const float inv = 1.f / sum;
      for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
        float xv[8], ov[8];
        load8(r + v * 8, xv);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int col = v * 8 + j;
          float p = __expf(xv[j] - mx) * inv;
          ov[j] = (p - (col == tgt ? 1.f : 0.f)) * grad_scale;
        }
        store8(r + v * 8, ov);
      }
      for (int j = tail0 + threadIdx.x; j < vocab; j += blockDim.x) {
        float p = __expf(bf2f(r[j]) - mx) / sum;
        r[j] = f2bf((p - (j == tgt ? 1.f : 0.f)) * grad_scale);
      }
    }
    __syncthreads();
  }
}

void cross_entropy_launch(void* logits, const void* targets, void* losses,
                          long long n_rows, int vocab, int ignore_index,
                          float grad_scale, int compute_grad,
                          hipStream_t stream) {
  int grid = n_rows < 2048 ? (int)n_rows : 2048;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(cross_entropy_fwd_bwd_kernel, dim3(grid), dim3(512), 0,
                     stream, (short*)logits, (const int*)targets,
                     (float*)losses, n_rows, vocab, ignore_index, grad_scale,
                     compute_grad);
}

}  // extern "C"
