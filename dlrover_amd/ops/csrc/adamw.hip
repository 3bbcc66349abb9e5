// Fused AdamW with fp32 master weights and bf16 model weights.
//
// One kernel updates master fp32 param, exp_avg, exp_avg_sq AND writes the
// bf16 working copy — four tensors touched once (vs ~10 passes for an
// unfused eager optimizer). Grads may be bf16 (post-DDP-allreduce) or fp32.
// The Python optimizer launches one call per parameter inside a hipGraph
// capture so the whole optimizer step replays as one graph launch
// (MI355X-native replacement for the reference's delegated Apex/Megatron
// fused Adam; BASELINE.json north star names fused Adam explicitly).
#include "kern_common.h"

typedef __attribute__((ext_vector_type(4))) float f32x4;

template <bool GRAD_BF16>
__global__ void adamw_kernel(
    float* __restrict__ param, const void* __restrict__ grad_in,
    float* __restrict__ exp_avg, float* __restrict__ exp_avg_sq,
    short* __restrict__ param_bf16, long long numel, float lr, float beta1,
    float beta2, float eps, float weight_decay, float bc1, float bc2,
    float grad_scale) {
  // 8 elements (2 independent b128 chains per array) per thread-iteration:
  // one chain alone leaves the memory pipes underfed on HBM3E — this loop is
  // pure bandwidth (28 B/param) and needs the outstanding-load depth
  const long long vecs = numel >> 3;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < vecs; i += (long long)gridDim.x * blockDim.x) {
    f32x4 p[2], m[2], v[2];
    float g[8];
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      p[h] = *reinterpret_cast<f32x4*>(param + i * 8 + h * 4);
      m[h] = *reinterpret_cast<f32x4*>(exp_avg + i * 8 + h * 4);
      v[h] = *reinterpret_cast<f32x4*>(exp_avg_sq + i * 8 + h * 4);
    }
    if (GRAD_BF16) {
      short8_t gv = *reinterpret_cast<const short8_t*>(
          (const short*)grad_in + i * 8);
#pragma unroll
      for (int j = 0; j < 8; ++j) g[j] = bf2f(gv[j]) * grad_scale;
    } else {
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        f32x4 gv =
            *reinterpret_cast<const f32x4*>((const float*)grad_in + i * 8 + h * 4);
#pragma unroll
        for (int j = 0; j < 4; ++j) g[h * 4 + j] = gv[j] * grad_scale;
      }
    }
    short8_t pb;
#pragma unroll
    for (int h = 0; h < 2; ++h) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float gj = g[h * 4 + j];
        m[h][j] = beta1 * m[h][j] + (1.f - beta1) * gj;
        v[h][j] = beta2 * v[h][j] + (1.f - beta2) * gj * gj;
        const float mhat = m[h][j] * bc1;
        const float vhat = v[h][j] * bc2;
        p[h][j] -= lr * (mhat / (sqrtf(vhat) + eps) + weight_decay * p[h][j]);
        pb[h * 4 + j] = f2bf(p[h][j]);
      }
    }
#pragma unroll
    for (int h = 0; h < 2; ++h) {
      *reinterpret_cast<f32x4*>(param + i * 8 + h * 4) = p[h];
      *reinterpret_cast<f32x4*>(exp_avg + i * 8 + h * 4) = m[h];
      *reinterpret_cast<f32x4*>(exp_avg_sq + i * 8 + h * 4) = v[h];
    }
    if (param_bf16)
      *reinterpret_cast<short8_t*>(param_bf16 + i * 8) = pb;
  }
  // scalar tail (numel % 8)
  const long long tail0 = vecs * 8;
  for (long long i = tail0 + (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < numel; i += (long long)gridDim.x * blockDim.x) {
    float g = GRAD_BF16 ? bf2f(((const short*)grad_in)[i])
                        : ((const float*)grad_in)[i];
    g *= grad_scale;
    float m = beta1 * exp_avg[i] + (1.f - beta1) * g;
    float v = beta2 * exp_avg_sq[i] + (1.f - beta2) * g * g;
    float p = param[i];
    p -= lr * ((m * bc1) / (sqrtf(v * bc2) + eps) + weight_decay * p);
    param[i] = p;
    exp_avg[i] = m;
    exp_avg_sq[i] = v;
    if (param_bf16) param_bf16[i] = f2bf(p);
  }
}

extern "C" void adamw_launch(void* param, const void* grad, void* exp_avg,
                  void* exp_avg_sq, void* param_bf16, long long numel,
                  float lr, float beta1, float beta2, float eps,
                  float weight_decay, int step, int grad_is_bf16,
                  float grad_scale, hipStream_t stream) {
  const float bc1 = 1.f / (1.f - powf(beta1, (float)step));
  const float bc2 = 1.f / (1.f - powf(beta2, (float)step));
  const int grid = grid_capped(numel >> 3, 256);
  if (grad_is_bf16) {
    hipLaunchKernelGGL((adamw_kernel<true>), dim3(grid), dim3(256), 0, stream,
                       (float*)param, grad, (float*)exp_avg,
                       (float*)exp_avg_sq, (short*)param_bf16, numel, lr,
                       beta1, beta2, eps, weight_decay, bc1, bc2, grad_scale);
  } else {
    hipLaunchKernelGGL((adamw_kernel<false>), dim3(grid), dim3(256), 0, stream,
                       (float*)param, grad, (float*)exp_avg,
                       (float*)exp_avg_sq, (short*)param_bf16, numel, lr,
                       beta1, beta2, eps, weight_decay, bc1, bc2, grad_scale);
  }
}
