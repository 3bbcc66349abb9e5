// Fused RMSNorm forward/backward for bf16 rows, fp32 accumulation.
//
// The reference delegates normalisation to Megatron/HF —
// this is part of the MI355X-native hot path (BASELINE.json north star).
// Memory-bound: target HBM ceiling via short8 vector loads (guide G13).
//
// Layout: x [N, H] bf16 row-major, w [H] bf16, y [N, H] bf16,
//         invrms [N] fp32 saved for backward.
#include "kern_common.h"

extern "C" {

// one block per row (grid-stride over rows), 256 threads, 8 elems/thread/iter
__global__ void rmsnorm_fwd_kernel(
    const short* __restrict__ x, const short* __restrict__ w,
    short* __restrict__ y, float* __restrict__ invrms,
    int n_rows, int hidden, float eps) {
  __shared__ float scratch[16];
  const int vecs = hidden >> 3;  // hidden % 8 == 0 enforced host-side
  for (int row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const short* xr = x + (long long)row * hidden;
    short* yr = y + (long long)row * hidden;
    float ssq = 0.f;
    for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
      float xv[8];
      load8(xr + v * 8, xv);
#pragma unroll
      for (int j = 0; j < 8; ++j) ssq += xv[j] * xv[j];
    }
    ssq = block_reduce_sum(ssq, scratch);
    const float inv = rsqrtf(ssq / hidden + eps);
    if (threadIdx.x == 0 && invrms) invrms[row] = inv;
    for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
      float xv[8], wv[8];
      load8(xr + v * 8, xv);
      load8(w + v * 8, wv);
      float out[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) out[j] = xv[j] * inv * wv[j];
      store8(yr + v * 8, out);
    }
    __syncthreads();
  }
}

// backward:
//   dx = inv * w * dy - x * inv^3 / H * sum_j(dy_j * w_j * x_j)
//   dw = sum_rows(dy * x * inv)
//
// dw accumulates in REGISTERS across the block's grid-stride rows (thread t
// always owns the same column set), flushed ONCE per block into fp32
// partial slices — profiling showed the per-element global atomicAdd version
// ran at 0.65 TB/s (profiles/r01_small1b_kernel_stats.txt); this removes
// n_rows*H atomics down to gridDim*H.
#define RMSN_MAX_VPT 8  // vecs/thread: supports hidden up to 8*8*256 = 16384
#define RMSN_CACHE_VPT 2  // dy/x register-cached when vecs <= 2*blockDim

}  // extern "C" — the templated bwd kernel needs C++ linkage

// CACHE=true (hidden <= 8*2*256 = 4096, the Llama case): pass 1's dy/x loads
// are kept in registers so pass 2 issues NO global reads except dh_extra —
// the old re-read version measured 257 us vs a ~40 us traffic bound
// (profiles/r01g). Every block owns its own dw_partial slice (n_partials ==
// gridDim, set host-side) and fully overwrites it with plain stores: the old
// 64-slice atomicAdd flush was ~8M contended RMW ops per call.
template <bool CACHE>
__global__ __launch_bounds__(256) void rmsnorm_bwd_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    const short* __restrict__ w, const float* __restrict__ invrms,
    const short* __restrict__ dh_extra,  // nullable: fused-add path's
                                         // residual-stream gradient
    short* __restrict__ dx, float* __restrict__ dw_partial,
    int n_rows, int hidden, int n_partials) {
  // ALL per-thread arrays are indexed only by compile-time unrolled vi —
  // runtime indexing would demote them to scratch (guide rule #20), which
  // is what held the previous version to ~1.4 TB/s
  constexpr int BD = 256;
  constexpr int VPT = CACHE ? RMSN_CACHE_VPT : RMSN_MAX_VPT;
  __shared__ float scratch[16];
  const int vecs = hidden >> 3;
  float* dwp = dw_partial + (long long)blockIdx.x * hidden;
  float dw_acc[VPT][8];
#pragma unroll
  for (int i = 0; i < VPT; ++i)
#pragma unroll
    for (int j = 0; j < 8; ++j) dw_acc[i][j] = 0.f;

  float dy_c[CACHE ? RMSN_CACHE_VPT : 1][8];
  float x_c[CACHE ? RMSN_CACHE_VPT : 1][8];

  for (int row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const short* dyr = dy + (long long)row * hidden;
    const short* xr = x + (long long)row * hidden;
    short* dxr = dx + (long long)row * hidden;
    const float inv = invrms[row];
    // pass 1: dot = sum(dy * w * x)
    float dot = 0.f;
#pragma unroll
    for (int vi = 0; vi < VPT; ++vi) {
      const int v = threadIdx.x + vi * BD;
      if (v < vecs) {
        float dyv[8], wv[8], xv[8];
        load8(dyr + v * 8, dyv);
        load8(w + v * 8, wv);
        load8(xr + v * 8, xv);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          dot += dyv[j] * wv[j] * xv[j];
          if (CACHE) {
            dy_c[vi][j] = dyv[j];
            x_c[vi][j] = xv[j];
          }
        }
      }
    }
    dot = block_reduce_sum(dot, scratch);
    const float k = dot * inv * inv * inv / hidden;
    // pass 2: dx + register dw accumulation (no dy/x re-read when CACHE)
#pragma unroll
    for (int vi = 0; vi < VPT; ++vi) {
      const int v = threadIdx.x + vi * BD;
      if (v < vecs) {
        float dyv[8], wv[8], xv[8], out[8];
        if (CACHE) {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            dyv[j] = dy_c[vi][j];
            xv[j] = x_c[vi][j];
          }
        } else {
          load8(dyr + v * 8, dyv);
          load8(xr + v * 8, xv);
        }
        load8(w + v * 8, wv);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          out[j] = dyv[j] * wv[j] * inv - xv[j] * k;
          dw_acc[vi][j] += dyv[j] * xv[j] * inv;
        }
        if (dh_extra) {
          float dhv[8];
          load8(dh_extra + (long long)row * hidden + v * 8, dhv);
#pragma unroll
          for (int j = 0; j < 8; ++j) out[j] += dhv[j];
        }
        store8(dxr + v * 8, out);
      }
    }
    __syncthreads();
  }
  // flush: this block's private slice, plain b128 stores
#pragma unroll
  for (int vi = 0; vi < VPT; ++vi) {
    const int v = threadIdx.x + vi * BD;
    if (v < vecs) {
      float4_t* dst = reinterpret_cast<float4_t*>(dwp + v * 8);
      dst[0] = {dw_acc[vi][0], dw_acc[vi][1], dw_acc[vi][2], dw_acc[vi][3]};
      dst[1] = {dw_acc[vi][4], dw_acc[vi][5], dw_acc[vi][6], dw_acc[vi][7]};
    }
  }
}

// stage 1: [n_partials, H] fp32 -> dw_f32 [H] via <=16 atomics per column
// (blockIdx.y picks a partial range; dw_f32 is zeroed host-side)
__global__ void reduce_partials_kernel(
    const float* __restrict__ partials, float* __restrict__ dw_f32,
    int n_partials, int hidden) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col >= hidden) return;
  const int per = (n_partials + gridDim.y - 1) / gridDim.y;
  const int p0 = blockIdx.y * per;
  const int p1 = min(p0 + per, n_partials);
  float acc = 0.f;
  for (int p = p0; p < p1; ++p)
    acc += partials[(long long)p * hidden + col];
  if (gridDim.y == 1)
    dw_f32[col] = acc;
  else
    atomicAdd(&dw_f32[col], acc);
}

// stage 2: fp32 -> bf16
__global__ void dw_cast_kernel(const float* __restrict__ dw_f32,
                               short* __restrict__ out, int hidden) {
  const int col = blockIdx.x * blockDim.x + threadIdx.x;
  if (col < hidden) out[col] = f2bf(dw_f32[col]);
}

// Fused residual-add + RMSNorm: h = x + resid; y = h * invrms(h) * w.
// Saves a full HBM round-trip of the eager residual add (4.7% of the 8B
// step as at::add — profiles/r01c). Writes BOTH h (needed as the next
// residual stream) and y.
__global__ void rmsnorm_add_fwd_kernel(
    const short* __restrict__ x, const short* __restrict__ resid,
    const short* __restrict__ w, short* __restrict__ h, short* __restrict__ y,
    float* __restrict__ invrms, int n_rows, int hidden, float eps) {
  __shared__ float scratch[16];
  const int vecs = hidden >> 3;
  for (int row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const short* xr = x + (long long)row * hidden;
    const short* rr = resid + (long long)row * hidden;
    short* hr = h + (long long)row * hidden;
    short* yr = y + (long long)row * hidden;
    float ssq = 0.f;
    for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
      float xv[8], rv[8], hv[8];
      load8(xr + v * 8, xv);
      load8(rr + v * 8, rv);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        hv[j] = xv[j] + rv[j];
        ssq += hv[j] * hv[j];
      }
      store8(hr + v * 8, hv);
    }
    ssq = block_reduce_sum(ssq, scratch);
    const float inv = rsqrtf(ssq / hidden + eps);
    if (threadIdx.x == 0 && invrms) invrms[row] = inv;
    for (int v = threadIdx.x; v < vecs; v += blockDim.x) {
      float hv[8], wv[8], out[8];
      load8(hr + v * 8, hv);
      load8(w + v * 8, wv);
#pragma unroll
      for (int j = 0; j < 8; ++j) out[j] = hv[j] * inv * wv[j];
      store8(yr + v * 8, out);
    }
    __syncthreads();
  }
}

extern "C" void rmsnorm_add_fwd_launch(const void* x, const void* resid,
                                       const void* w, void* h, void* y,
                                       void* invrms, int n_rows, int hidden,
                                       float eps, hipStream_t stream) {
  int grid = n_rows < 2048 ? n_rows : 2048;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(rmsnorm_add_fwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const short*)x, (const short*)resid, (const short*)w,
                     (short*)h, (short*)y, (float*)invrms, n_rows, hidden,
                     eps);
}

extern "C" void rmsnorm_fwd_launch(const void* x, const void* w, void* y,
                                   void* invrms, int n_rows, int hidden,
                                   float eps, hipStream_t stream) {
  int grid = n_rows < 2048 ? n_rows : 2048;
  if (grid < 1) grid = 1;
  hipLaunchKernelGGL(rmsnorm_fwd_kernel, dim3(grid), dim3(256), 0, stream,
                     (const short*)x, (const short*)w, (short*)y,
                     (float*)invrms, n_rows, hidden, eps);
}

extern "C" {

// n_partials MUST equal the launch grid (each block owns one slice);
// dw_f32 is a zero-init [hidden] fp32 scratch for the two-stage reduce.
void rmsnorm_bwd_launch(const void* dy, const void* x, const void* w,
                        const void* invrms, const void* dh_extra, void* dx,
                        void* dw_partial, void* dw_f32, void* dw, int n_rows,
                        int hidden, int n_partials, hipStream_t stream) {
  int grid = n_rows < 2048 ? n_rows : 2048;
  if (grid < 1) grid = 1;
  const int vecs = hidden >> 3;
  if (vecs <= RMSN_CACHE_VPT * 256) {
    hipLaunchKernelGGL((rmsnorm_bwd_kernel<true>), dim3(grid), dim3(256), 0,
                       stream, (const short*)dy, (const short*)x,
                       (const short*)w, (const float*)invrms,
                       (const short*)dh_extra, (short*)dx, (float*)dw_partial,
                       n_rows, hidden, n_partials);
  } else {
    hipLaunchKernelGGL((rmsnorm_bwd_kernel<false>), dim3(grid), dim3(256), 0,
                       stream, (const short*)dy, (const short*)x,
                       (const short*)w, (const float*)invrms,
                       (const short*)dh_extra, (short*)dx, (float*)dw_partial,
                       n_rows, hidden, n_partials);
  }
  // enough y-blocks to fill the chip: 16 x-blocks alone leave 240 CUs idle
  // (measured 64 us/call at ny<=16 — latency-bound on 16 KB-strided reads)
  int ny = (n_partials + 31) / 32;
  if (ny > 64) ny = 64;
  if (ny < 1) ny = 1;
  hipLaunchKernelGGL(reduce_partials_kernel,
                     dim3((hidden + 255) / 256, ny), dim3(256), 0, stream,
                     (const float*)dw_partial, (float*)dw_f32, n_partials,
                     hidden);
  hipLaunchKernelGGL(dw_cast_kernel, dim3((hidden + 255) / 256), dim3(256), 0,
                     stream, (const float*)dw_f32, (short*)dw, hidden);
}

}  // extern "C"
