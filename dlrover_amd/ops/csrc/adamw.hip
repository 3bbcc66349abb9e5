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
  const long long vecs = numel >> 2;
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       i < vecs; i += (long long)gridDim.x * blockDim.x) {
    f32x4 p = *reinterpret_cast<f32x4*>(param + i * 4);
    f32x4 m = *reinterpret_cast<f32x4*>(exp_avg + i * 4);
    f32x4 v = *reinterpret_cast<f32x4*>(exp_avg_sq + i * 4);
    float g[4];
    if (GRAD_BF16) {
      short4_t gv = *reinterpret_cast<const short4_t*>(
          (const short*)grad_in + i * 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) g[j] = bf2f(gv[j]) * grad_scale;
    } else {
      f32x4 gv = *reinterpret_cast<const f32x4*>((const float*)grad_in + i * 4);
#pragma unroll
      for (int j = 0; j < 4; ++j) g[j] = gv[j] * grad_scale;
    }
    short4_t pb;
#pragma unroll
    for (int j = 0; j < 4; ++j) {
      m[j] = beta1 * m[j] + (1.f - beta1) * g[j];
      v[j] = beta2 * v[j] + (1.f - beta2) * g[j] * g[j];
      const float mhat = m[j] * bc1;
      const float vhat = v[j] * bc2;
      p[j] -= lr * (mhat / (sqrtf(vhat) + eps) + weight_decay * p[j]);
      pb[j] = f2bf(p[j]);
    }
    *reinterpret_cast<f32x4*>(param + i * 4) = p;
    *reinterpret_cast<f32x4*>(exp_avg + i * 4) = m;
    *reinterpret_cast<f32x4*>(exp_avg_sq + i * 4) = v;
    if (param_bf16) *reinterpret_cast<short4_t*>(param_bf16 + i * 4) = pb;
  }
  // scalar tail (numel % 4)
  const long long tail0 = vecs * 4;
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
  const int grid = grid_capped(numel >> 2, 256);
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
