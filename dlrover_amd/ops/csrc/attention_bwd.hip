// Flash attention backward for MI355X (bf16, causal, GQA, D=128).
//
// FlashAttention-2 style recompute backward, two kernels + a preprocess:
//   fa_bwd_pre:  Dvec[b,h,s] = rowsum(dO ⊙ O)
//   fa_bwd_dq:   block per (b,h,q-tile): recompute P from saved logsumexp,
//                dP = dO @ V^T, dS = P ⊙ (dP − Dvec) · scale, dQ += dS @ K
//   fa_bwd_dkv:  block per (b,h,kv-tile): transposed recompute
//                P^T = exp(K Q^T·scale − L), dV += P^T @ dO,
//                dP^T = V @ dO^T, dS^T = P^T ⊙ (dP^T − Dvec)·scale,
//                dK += dS^T @ Q; GQA head groups accumulate into fp32
//                dK/dV buffers with device atomics (cast to bf16 after).
//
// Same MFMA fragment mappings and LDS XOR swizzle as attention.hip
// (hardware-verified by tests/test_mfma_gpu.py).
#include "kern_common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define FA_D 128
#define FA_T 64  // tile rows (both q and kv)

__device__ __forceinline__ int swz2(int row, int byte_col) {
  return byte_col ^ ((row & 7) << 4);
}

// B-fragment read from an LDS buffer laid out [n_rows][k_cols] bf16 where the
// fragment wants B[k][n] = buf[n][k]: lane reads row n0+sub, 8 contiguous k.
__device__ __forceinline__ bf16x8 ld_bT(const char* lds, int n0, int sub,
                                        int k0, int quarter, int row_bytes) {
  const int row = n0 + sub;
  const int bc = (k0 + quarter * 8) * 2;
  return *reinterpret_cast<const bf16x8*>(lds + row * row_bytes + swz2(row, bc));
}

// A-fragment read: lane reads row sub, 8 contiguous k at k0+quarter*8.
__device__ __forceinline__ bf16x8 ld_a(const char* lds, int sub, int k0,
                                       int quarter, int row_bytes) {
  const int bc = (k0 + quarter * 8) * 2;
  return *reinterpret_cast<const bf16x8*>(lds + sub * row_bytes + swz2(sub, bc));
}

// B-fragment read straight from GLOBAL memory: B[k][n] = src[n][k] where src
// rows are row_stride apart. The 64x128 tile is L1/L2-resident across the
// 16 fragment reads per iteration, so skipping the LDS copy trades a little
// cache traffic for 32 KB of LDS (occupancy: 1 -> 3 waves/SIMD in dkv).
__device__ __forceinline__ bf16x8 ld_bT_global(const short* src,
                                               long long row_stride, int n0,
                                               int sub, int k0, int quarter) {
  return *reinterpret_cast<const bf16x8*>(
      src + (long long)(n0 + sub) * row_stride + k0 + quarter * 8);
}

// stage a [64][128] bf16 global tile into LDS row-major (swizzled), and
// optionally also transposed into a [128][64] buffer.
__device__ __forceinline__ void stage_tile(const short* __restrict__ src,
                                           long long row_stride, char* row_lds,
                                           char* tr_lds, int tid) {
#pragma unroll
  for (int c = 0; c < 4; ++c) {
    const int linear = (tid * 4 + c) * 8;
    const int row = linear / FA_D;
    const int col = linear % FA_D;
    bf16x8 v8 =
        *reinterpret_cast<const bf16x8*>(src + (long long)row * row_stride + col);
    if (row_lds)
      *reinterpret_cast<bf16x8*>(row_lds + row * 256 + swz2(row, col * 2)) = v8;
    if (tr_lds) {
#pragma unroll
      for (int i = 0; i < 8; ++i)
        *reinterpret_cast<__bf16*>(
            tr_lds + (col + i) * 128 + swz2(col + i, row * 2)) = v8[i];
    }
  }
}

// ---------------------------------------------------------------------------
// preprocess: Dvec = rowsum(dO * O), one 64-lane wave per 2 rows
// ---------------------------------------------------------------------------

__global__ void fa_bwd_pre_kernel(const short* __restrict__ dout,
                                  const short* __restrict__ out,
                                  float* __restrict__ dvec, int B, int H,
                                  int S, long long os_b, long long os_h,
                                  long long os_s, long long ds_b,
                                  long long ds_h, long long ds_s) {
  const int lane32 = threadIdx.x & 31;
  const long long group0 =
      ((long long)blockIdx.x * blockDim.x + threadIdx.x) >> 5;
  const long long stride = ((long long)gridDim.x * blockDim.x) >> 5;
  const long long rows = (long long)B * H * S;
  for (long long row = group0; row < rows; row += stride) {
    const int b = (int)(row / ((long long)H * S));
    const int h = (int)((row / S) % H);
    const int sI = (int)(row % S);
    const short* dp =
        dout + b * ds_b + h * ds_h + (long long)sI * ds_s + lane32 * 4;
    const short* op =
        out + b * os_b + h * os_h + (long long)sI * os_s + lane32 * 4;
    float acc = 0.f;
    short4_t d4 = *reinterpret_cast<const short4_t*>(dp);
    short4_t o4 = *reinterpret_cast<const short4_t*>(op);
#pragma unroll
    for (int i = 0; i < 4; ++i) acc += bf2f(d4[i]) * bf2f(o4[i]);
    // reduce across the 32 lanes sharing this row (fits inside a half-wave)
#pragma unroll
    for (int s = 16; s > 0; s >>= 1) acc += __shfl_xor(acc, s, 64);
    if (lane32 == 0) dvec[row] = acc;
  }
}

// ---------------------------------------------------------------------------
// dQ kernel
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void fa_bwd_dq_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ dvec,
    short* __restrict__ dq, int B, int H, int HKV, int S, float scale,
    long long qs_b, long long qs_h, long long qs_s,
    long long ks_b, long long ks_h, long long ks_s,
    long long vs_b, long long vs_h, long long vs_s,
    long long ds_b, long long ds_h, long long ds_s) {
  __shared__ char k_lds[FA_T * 256];    // K row-major [64][128]
  __shared__ char kt_lds[FA_D * 128];   // K^T [128][64]
  __shared__ char v_lds[FA_T * 256];    // V row-major [64][128]
  __shared__ char ds_lds[4 * 16 * 128]; // per-wave dS [16][64] bf16

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int quarter = lane >> 4;
  const int sub = lane & 15;

  const int qt = blockIdx.x, h = blockIdx.y, b = blockIdx.z;
  const int g = h / (H / HKV);
  // dout shares q's layout (torch grads of a permuted view may differ, the
  // binding normalizes); q/dout/dq all use the q-strides triple here
  const short* q_blk = q + (long long)b * qs_b + (long long)h * qs_h +
                       (long long)(qt * FA_T) * qs_s;
  const short* do_blk = dout + (long long)b * ds_b + (long long)h * ds_h +
                        (long long)(qt * FA_T) * ds_s;
  const short* k_head = k + (long long)b * ks_b + (long long)g * ks_h;
  const short* v_head = v + (long long)b * vs_b + (long long)g * vs_h;
  const long long row_base = ((long long)b * H + h) * S + qt * FA_T;
  const int qrow_w = wave * 16;

  // persistent A fragments: Q band and dO band
  bf16x8 aq[4], ado[4];
#pragma unroll
  for (int ks = 0; ks < 4; ++ks) {
    const int ko = ks * 32 + quarter * 8;
    aq[ks] = *reinterpret_cast<const bf16x8*>(
        q_blk + (long long)(qrow_w + sub) * qs_s + ko);
    ado[ks] = *reinterpret_cast<const bf16x8*>(
        do_blk + (long long)(qrow_w + sub) * ds_s + ko);
  }
  // per-row stats for this lane's 4 C-rows
  float lse_r[4], dv_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const long long rr = row_base + qrow_w + quarter * 4 + r;
    lse_r[r] = lse[rr];
    dv_r[r] = dvec[rr];
  }

  f32x4_t acc_dq[8];
#pragma unroll
  for (int n = 0; n < 8; ++n) acc_dq[n] = {0.f, 0.f, 0.f, 0.f};
  char* ds_wave = ds_lds + wave * 16 * 128;

  for (int kt = 0; kt <= qt; ++kt) {
    stage_tile(k_head + (long long)(kt * FA_T) * ks_s, ks_s, k_lds, kt_lds, tid);
    stage_tile(v_head + (long long)(kt * FA_T) * vs_s, vs_s, v_lds, nullptr, tid);
    __syncthreads();

    // S = Q @ K^T ; dP = dO @ V^T  (both [16,64])
    f32x4_t acc_s[4], acc_dp[4];
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      acc_s[n] = {0.f, 0.f, 0.f, 0.f};
      acc_dp[n] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8 bk = ld_bT(k_lds, n * 16, sub, ks * 32, quarter, 256);
        acc_s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[ks], bk,
                                                           acc_s[n], 0, 0, 0);
        bf16x8 bv = ld_bT(v_lds, n * 16, sub, ks * 32, quarter, 256);
        acc_dp[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ado[ks], bv,
                                                            acc_dp[n], 0, 0, 0);
      }
    }
    // dS = P * (dP - Dvec) * scale, staged for the next MFMA's A operand
    const int row_glob0 = qt * FA_T + qrow_w + quarter * 4;
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const int col_glob = kt * FA_T + n * 16 + sub;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = __expf(acc_s[n][r] * scale - lse_r[r]);
        if (col_glob > row_glob0 + r) p = 0.f;
        const float ds = p * (acc_dp[n][r] - dv_r[r]) * scale;
        const int prow = quarter * 4 + r;
        const int pcol = n * 16 + sub;
        *reinterpret_cast<__bf16*>(ds_wave + prow * 128 + swz2(prow, pcol * 2)) =
            (__bf16)ds;
      }
    }
    __syncthreads();
    // dQ += dS @ K : B[k=kv][n=dim] = KT[n][k]
#pragma unroll
    for (int n = 0; n < 8; ++n) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16x8 a = ld_a(ds_wave, sub, ks * 32, quarter, 128);
        bf16x8 bb = ld_bT(kt_lds, n * 16, sub, ks * 32, quarter, 128);
        acc_dq[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bb, acc_dq[n],
                                                            0, 0, 0);
      }
    }
    __syncthreads();
  }

  short* dq_blk = dq + (long long)b * qs_b + (long long)h * qs_h +
                  (long long)(qt * FA_T) * qs_s;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int orow = quarter * 4 + r;
#pragma unroll
    for (int n = 0; n < 8; ++n)
      dq_blk[((long long)qrow_w + orow) * qs_s + n * 16 + sub] =
          f2bf(acc_dq[n][r]);
  }
}

// ---------------------------------------------------------------------------
// dK/dV kernel (transposed recompute; atomics over GQA head groups)
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(256) void fa_bwd_dkv_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ dvec,
    float* __restrict__ dk32, float* __restrict__ dv32, int B, int H,
    int HKV, int S, float scale,
    long long qs_b, long long qs_h, long long qs_s,
    long long ks_b, long long ks_h, long long ks_s,
    long long vs_b, long long vs_h, long long vs_s,
    long long ds_b, long long ds_h, long long ds_s) {
  // measured: LDS-staged row-major Q/dO beats global-cached fragment reads
  // (1992 vs 2260 us/call) even at occupancy 1 — keep them in LDS
  __shared__ char q_lds[FA_T * 256];     // Q row-major
  __shared__ char qt_lds[FA_D * 128];    // Q^T
  __shared__ char do_lds[FA_T * 256];    // dO row-major
  __shared__ char dot_lds[FA_D * 128];   // dO^T
  __shared__ char pt_lds[4 * 16 * 128];  // per-wave P^T [16kv][64q]
  __shared__ char dst_lds[4 * 16 * 128]; // per-wave dS^T [16kv][64q]

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int quarter = lane >> 4;
  const int sub = lane & 15;

  const int kt = blockIdx.x, h = blockIdx.y, b = blockIdx.z;
  const int g = h / (H / HKV);
  const short* k_blk = k + (long long)b * ks_b + (long long)g * ks_h +
                       (long long)(kt * FA_T) * ks_s;
  const short* v_blk = v + (long long)b * vs_b + (long long)g * vs_h +
                       (long long)(kt * FA_T) * vs_s;
  const short* q_head = q + (long long)b * qs_b + (long long)h * qs_h;
  const short* do_head = dout + (long long)b * ds_b + (long long)h * ds_h;
  const long long row_base = ((long long)b * H + h) * S;
  const int kvrow_w = wave * 16;

  // persistent A fragments: K band and V band (this wave's 16 kv rows)
  bf16x8 ak[4], av[4];
#pragma unroll
  for (int ks = 0; ks < 4; ++ks) {
    const int ko = ks * 32 + quarter * 8;
    ak[ks] = *reinterpret_cast<const bf16x8*>(
        k_blk + (long long)(kvrow_w + sub) * ks_s + ko);
    av[ks] = *reinterpret_cast<const bf16x8*>(
        v_blk + (long long)(kvrow_w + sub) * vs_s + ko);
  }

  f32x4_t acc_dk[8], acc_dv[8];
#pragma unroll
  for (int n = 0; n < 8; ++n) {
    acc_dk[n] = {0.f, 0.f, 0.f, 0.f};
    acc_dv[n] = {0.f, 0.f, 0.f, 0.f};
  }
  char* pt_wave = pt_lds + wave * 16 * 128;
  char* dst_wave = dst_lds + wave * 16 * 128;

  const int n_qt = S / FA_T;
  for (int qt = kt; qt < n_qt; ++qt) {
    const short* q_tile = q_head + (long long)(qt * FA_T) * qs_s;
    const short* do_tile = do_head + (long long)(qt * FA_T) * ds_s;
    stage_tile(q_tile, qs_s, q_lds, qt_lds, tid);
    stage_tile(do_tile, ds_s, do_lds, dot_lds, tid);
    __syncthreads();

    // S^T = K_band @ Q^T ; dP^T = V_band @ dO^T   (both [16kv, 64q])
    f32x4_t acc_st[4], acc_dpt[4];
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      acc_st[n] = {0.f, 0.f, 0.f, 0.f};
      acc_dpt[n] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        // B[k=dim][n=q] = Q[n][k] (row-major Q) / dO[n][k]
        bf16x8 bq = ld_bT(q_lds, n * 16, sub, ks * 32, quarter, 256);
        acc_st[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak[ks], bq,
                                                            acc_st[n], 0, 0, 0);
        bf16x8 bdo = ld_bT(do_lds, n * 16, sub, ks * 32, quarter, 256);
        acc_dpt[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            av[ks], bdo, acc_dpt[n], 0, 0, 0);
      }
    }
    // P^T and dS^T (rows = kv, cols = q): stats indexed by the q COLUMN
    const int kvrow0 = kt * FA_T + kvrow_w + quarter * 4;
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const int qcol = qt * FA_T + n * 16 + sub;
      const float l_col = lse[row_base + qcol];
      const float d_col = dvec[row_base + qcol];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = __expf(acc_st[n][r] * scale - l_col);
        if (qcol < kvrow0 + r) p = 0.f;  // causal: q must be >= kv
        const float ds = p * (acc_dpt[n][r] - d_col) * scale;
        const int prow = quarter * 4 + r;
        const int pcol = n * 16 + sub;
        *reinterpret_cast<__bf16*>(pt_wave + prow * 128 + swz2(prow, pcol * 2)) =
            (__bf16)p;
        *reinterpret_cast<__bf16*>(
            dst_wave + prow * 128 + swz2(prow, pcol * 2)) = (__bf16)ds;
      }
    }
    __syncthreads();
    // dV += P^T @ dO : B[k=q][n=dim] = dOT[n][k]
    // dK += dS^T @ Q : B[k=q][n=dim] = QT[n][k]
#pragma unroll
    for (int n = 0; n < 8; ++n) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16x8 apt = ld_a(pt_wave, sub, ks * 32, quarter, 128);
        bf16x8 bdot = ld_bT(dot_lds, n * 16, sub, ks * 32, quarter, 128);
        acc_dv[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(apt, bdot,
                                                            acc_dv[n], 0, 0, 0);
        bf16x8 adst = ld_a(dst_wave, sub, ks * 32, quarter, 128);
        bf16x8 bqt = ld_bT(qt_lds, n * 16, sub, ks * 32, quarter, 128);
        acc_dk[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(adst, bqt,
                                                            acc_dk[n], 0, 0, 0);
      }
    }
    __syncthreads();
  }

  // accumulate into fp32 dK/dV (GQA head groups collide -> device atomics).
  // dk32/dv32 are CONTIGUOUS [B,HKV,S,D] buffers allocated by the binding.
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int krow = quarter * 4 + r;
    const long long out_off =
        (((long long)b * HKV + g) * S + kt * FA_T + kvrow_w + krow) * FA_D;
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      atomicAdd(&dk32[out_off + n * 16 + sub], acc_dk[n][r]);
      atomicAdd(&dv32[out_off + n * 16 + sub], acc_dv[n][r]);
    }
  }
}

__global__ void f32_to_bf16_kernel(const float* __restrict__ src,
                                   short* __restrict__ dst, long long n) {
  for (long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x; i < n;
       i += (long long)gridDim.x * blockDim.x)
    dst[i] = f2bf(src[i]);
}

// ============================================================================
// dKV v2: identical math to fa_bwd_dkv_kernel but the q dimension is staged
// in 32-row SUBTILES, halving every q-side LDS buffer: 80 KB -> 40 KB per
// block, so 2-4 blocks/CU instead of 1 (the v1 kernel is latency-exposed at
// occupancy 1 — 1393 us/call vs dq's 781 at similar work, profiles/r01g).
// MFMA count per q-row is unchanged; only barrier/staging rounds double.

#define FA_TS 32  // staged q rows per round

// 64-byte-row swizzle for the [128][32] transposed buffers (the (row&7)<<4
// key of swz2 would wrap outside a 64 B row)
__device__ __forceinline__ int swz64(int row, int byte_col) {
  return byte_col ^ ((row & 3) << 4);
}

__device__ __forceinline__ bf16x8 ld_bT64(const char* lds, int n0, int sub,
                                          int k0, int quarter) {
  const int row = n0 + sub;
  const int bc = (k0 + quarter * 8) * 2;
  return *reinterpret_cast<const bf16x8*>(lds + row * 64 + swz64(row, bc));
}

__device__ __forceinline__ bf16x8 ld_a64(const char* lds, int sub, int k0,
                                         int quarter) {
  const int bc = (k0 + quarter * 8) * 2;
  return *reinterpret_cast<const bf16x8*>(lds + sub * 64 + swz64(sub, bc));
}

// stage a [32][128] tile row-major (+ transposed [128][32])
__device__ __forceinline__ void stage_tile32(const short* __restrict__ src,
                                             long long row_stride,
                                             char* row_lds, char* tr_lds,
                                             int tid) {
#pragma unroll
  for (int c = 0; c < 2; ++c) {
    const int linear = (tid * 2 + c) * 8;
    const int row = linear / FA_D;
    const int col = linear % FA_D;
    bf16x8 v8 =
        *reinterpret_cast<const bf16x8*>(src + (long long)row * row_stride + col);
    *reinterpret_cast<bf16x8*>(row_lds + row * 256 + swz2(row, col * 2)) = v8;
#pragma unroll
    for (int i = 0; i < 8; ++i)
      *reinterpret_cast<__bf16*>(
          tr_lds + (col + i) * 64 + swz64(col + i, row * 2)) = v8[i];
  }
}

__global__ __launch_bounds__(256) void fa_bwd_dkv_v2_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ dvec,
    float* __restrict__ dk32, float* __restrict__ dv32, int B, int H,
    int HKV, int S, float scale,
    long long qs_b, long long qs_h, long long qs_s,
    long long ks_b, long long ks_h, long long ks_s,
    long long vs_b, long long vs_h, long long vs_s,
    long long ds_b, long long ds_h, long long ds_s) {
  __shared__ char q_lds[FA_TS * 256];      // Q rows (8 KB)
  __shared__ char qt_lds[FA_D * 64];       // Q^T [128][32] (8 KB)
  __shared__ char do_lds[FA_TS * 256];     // dO rows (8 KB)
  __shared__ char dot_lds[FA_D * 64];      // dO^T (8 KB)
  __shared__ char pt_lds[4 * 16 * 64];     // per-wave P^T [16kv][32q] (4 KB)
  __shared__ char dst_lds[4 * 16 * 64];    // per-wave dS^T (4 KB)

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int quarter = lane >> 4;
  const int sub = lane & 15;

  const int kt = blockIdx.x, h = blockIdx.y, b = blockIdx.z;
  const int g = h / (H / HKV);
  const short* k_blk = k + (long long)b * ks_b + (long long)g * ks_h +
                       (long long)(kt * FA_T) * ks_s;
  const short* v_blk = v + (long long)b * vs_b + (long long)g * vs_h +
                       (long long)(kt * FA_T) * vs_s;
  const short* q_head = q + (long long)b * qs_b + (long long)h * qs_h;
  const short* do_head = dout + (long long)b * ds_b + (long long)h * ds_h;
  const long long row_base = ((long long)b * H + h) * S;
  const int kvrow_w = wave * 16;

  bf16x8 ak[4], av[4];
#pragma unroll
  for (int ks = 0; ks < 4; ++ks) {
    const int ko = ks * 32 + quarter * 8;
    ak[ks] = *reinterpret_cast<const bf16x8*>(
        k_blk + (long long)(kvrow_w + sub) * ks_s + ko);
    av[ks] = *reinterpret_cast<const bf16x8*>(
        v_blk + (long long)(kvrow_w + sub) * vs_s + ko);
  }

  f32x4_t acc_dk[8], acc_dv[8];
#pragma unroll
  for (int n = 0; n < 8; ++n) {
    acc_dk[n] = {0.f, 0.f, 0.f, 0.f};
    acc_dv[n] = {0.f, 0.f, 0.f, 0.f};
  }
  char* pt_wave = pt_lds + wave * 16 * 64;
  char* dst_wave = dst_lds + wave * 16 * 64;

  for (int qs0 = kt * FA_T; qs0 < S; qs0 += FA_TS) {
    stage_tile32(q_head + (long long)qs0 * qs_s, qs_s, q_lds, qt_lds, tid);
    stage_tile32(do_head + (long long)qs0 * ds_s, ds_s, do_lds, dot_lds, tid);
    __syncthreads();

    // S^T = K_band @ Q^T ; dP^T = V_band @ dO^T   (both [16kv, 32q])
    f32x4_t acc_st[2], acc_dpt[2];
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      acc_st[n] = {0.f, 0.f, 0.f, 0.f};
      acc_dpt[n] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8 bq = ld_bT(q_lds, n * 16, sub, ks * 32, quarter, 256);
        acc_st[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ak[ks], bq,
                                                            acc_st[n], 0, 0, 0);
        bf16x8 bdo = ld_bT(do_lds, n * 16, sub, ks * 32, quarter, 256);
        acc_dpt[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            av[ks], bdo, acc_dpt[n], 0, 0, 0);
      }
    }
    const int kvrow0 = kt * FA_T + kvrow_w + quarter * 4;
#pragma unroll
    for (int n = 0; n < 2; ++n) {
      const int qcol = qs0 + n * 16 + sub;
      const float l_col = lse[row_base + qcol];
      const float d_col = dvec[row_base + qcol];
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float p = __expf(acc_st[n][r] * scale - l_col);
        if (qcol < kvrow0 + r) p = 0.f;
        const float ds = p * (acc_dpt[n][r] - d_col) * scale;
        const int prow = quarter * 4 + r;
        const int pcol = n * 16 + sub;
        *reinterpret_cast<__bf16*>(pt_wave + prow * 64 + swz64(prow, pcol * 2)) =
            (__bf16)p;
        *reinterpret_cast<__bf16*>(
            dst_wave + prow * 64 + swz64(prow, pcol * 2)) = (__bf16)ds;
      }
    }
    __syncthreads();
    // dV += P^T @ dO ; dK += dS^T @ Q   (k dimension = 32 staged q rows)
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      bf16x8 apt = ld_a64(pt_wave, sub, 0, quarter);
      bf16x8 bdot = ld_bT64(dot_lds, n * 16, sub, 0, quarter);
      acc_dv[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(apt, bdot,
                                                          acc_dv[n], 0, 0, 0);
      bf16x8 adst = ld_a64(dst_wave, sub, 0, quarter);
      bf16x8 bqt = ld_bT64(qt_lds, n * 16, sub, 0, quarter);
      acc_dk[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(adst, bqt,
                                                          acc_dk[n], 0, 0, 0);
    }
    __syncthreads();
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int krow = quarter * 4 + r;
    const long long out_off =
        (((long long)b * HKV + g) * S + kt * FA_T + kvrow_w + krow) * FA_D;
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      atomicAdd(&dk32[out_off + n * 16 + sub], acc_dk[n][r]);
      atomicAdd(&dv32[out_off + n * 16 + sub], acc_dv[n][r]);
    }
  }
}


extern "C" {

void fa_bwd_pre_launch(const void* dout, const void* out, void* dvec, int B,
                       int H, int S, const long long* ost,
                       const long long* dst, hipStream_t stream) {
  const long long threads = (long long)B * H * S * 32;
  hipLaunchKernelGGL(fa_bwd_pre_kernel, dim3(grid_capped(threads, 256)),
                     dim3(256), 0, stream, (const short*)dout,
                     (const short*)out, (float*)dvec, B, H, S, ost[0], ost[1],
                     ost[2], dst[0], dst[1], dst[2]);
}

void fa_bwd_dq_launch(const void* q, const void* k, const void* v,
                      const void* dout, const void* lse, const void* dvec,
                      void* dq, int B, int H, int HKV, int S, float scale,
                      const long long* st, hipStream_t stream) {
  dim3 grid(S / FA_T, H, B);
  hipLaunchKernelGGL(fa_bwd_dq_kernel, grid, dim3(256), 0, stream,
                     (const short*)q, (const short*)k, (const short*)v,
                     (const short*)dout, (const float*)lse,
                     (const float*)dvec, (short*)dq, B, H, HKV, S, scale,
                     st[0], st[1], st[2], st[3], st[4], st[5], st[6], st[7],
                     st[8], st[9], st[10], st[11]);
}

void fa_bwd_dkv_launch(const void* q, const void* k, const void* v,
                       const void* dout, const void* lse, const void* dvec,
                       void* dk32, void* dv32, int B, int H, int HKV, int S,
                       float scale, const long long* st, hipStream_t stream) {
  dim3 grid(S / FA_T, H, B);
  // v1 (64-row tiles, 80 KB LDS) measured FASTER than the 32-row-subtile
  // v2 (1394 vs 1515 us/call): the subtile variant stays VGPR-occupancy-
  // limited, so halving LDS bought nothing and the doubled staging/barrier
  // rounds cost ~9%. DLROVER_FA_BWD_V2=1 keeps v2 selectable for re-testing
  // after a register-pressure rework (round 2).
  static const bool use_v1 = []() {
    const char* e = getenv("DLROVER_FA_BWD_V2");
    return !(e != nullptr && e[0] == '1');
  }();
  if (use_v1) {
    hipLaunchKernelGGL(fa_bwd_dkv_kernel, grid, dim3(256), 0, stream,
                       (const short*)q, (const short*)k, (const short*)v,
                       (const short*)dout, (const float*)lse,
                       (const float*)dvec, (float*)dk32, (float*)dv32, B, H,
                       HKV, S, scale,
                       st[0], st[1], st[2], st[3], st[4], st[5], st[6], st[7],
                       st[8], st[9], st[10], st[11]);
  } else {
    hipLaunchKernelGGL(fa_bwd_dkv_v2_kernel, grid, dim3(256), 0, stream,
                       (const short*)q, (const short*)k, (const short*)v,
                       (const short*)dout, (const float*)lse,
                       (const float*)dvec, (float*)dk32, (float*)dv32, B, H,
                       HKV, S, scale,
                       st[0], st[1], st[2], st[3], st[4], st[5], st[6], st[7],
                       st[8], st[9], st[10], st[11]);
  }
}

void f32_to_bf16_launch(const void* src, void* dst, long long n,
                        hipStream_t stream) {
  hipLaunchKernelGGL(f32_to_bf16_kernel, dim3(grid_capped(n, 256)), dim3(256),
                     0, stream, (const float*)src, (short*)dst, n);
}

}  // extern "C"
