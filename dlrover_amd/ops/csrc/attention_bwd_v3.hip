// Flash attention backward V3 for MI355X — 32x32x16 MFMA swapped-operand
// rebuild of the round-1 kernels (the #1/#2 custom-kernel step-time
// consumers, profiles/r01h). Same structural ideas as attention_v3.hip:
//
// dQ kernel (q-major, per-lane q column):
//   S^T = K Q^T   (C col = q: per-lane scalar L, Dvec, causal mask)
//   P^T = exp(S^T*scale - L)
//   dP^T = V dO^T (B operand dO^T is the lane's OWN dO row — register frags)
//   dS^T = P^T .* (dP^T - Dvec) * scale   (all in-lane)
//   dQ^T += K^T dS^T  (A = K^T via tr_b16 cooperative gather from row-major
//                      K LDS; B = dS^T via the pair-swap fragment assembly)
//
// dKV kernel (kv-major, per-lane kv column):
//   S = Q K^T     (C col = kv; L/Dvec per q ROW read from a small LDS tile)
//   P = exp(S*scale - L)
//   dV^T += dO^T P    (A = dO^T tr-gathered; B = P pair-swapped)
//   dP = dO V^T       (B operand V^T is the lane's OWN V row — registers)
//   dS = P .* (dP - Dvec) * scale
//   dK^T += Q^T dS    (A = Q^T tr-gathered; B = dS pair-swapped)
//   GQA: the rep q-heads of the kv group iterate INSIDE the block; work is
//   chunked over (head, q-tile) pairs across gridDim.y for occupancy at
//   small B*HKV, accumulating into fp32 dK/dV buffers with atomics only
//   when chunked.
//
// Fragment layouts + tr_b16 gather semantics: probe-verified
// (tests/test_mfma_gpu.py, profiles/r02a_fa_v3.txt).
#include <cstdlib>
#include "kern_common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;
typedef __attribute__((ext_vector_type(2))) unsigned int u32x2_t;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4_t;
typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

#define FB_D 128
#define FB_QB 256  // q rows per dq block / kv rows per dkv block
#define FB_T 64    // staged tile rows

__device__ __forceinline__ int swzb(int row, int byte_col) {
  return byte_col ^ ((row & 7) << 4);
}

__device__ __forceinline__ unsigned pk_bf16(float lo, float hi) {
  bf16x2 v = {(__bf16)lo, (__bf16)hi};
  return *reinterpret_cast<unsigned*>(&v);
}

// cooperative-transpose A-fragment: X^T[d=n*32+(l&31)][rb+half*8 .. +8] from
// a row-major swizzled [rows][128] LDS tile (see attention_v3.hip)
__device__ __forceinline__ bf16x8 trT_frag(const char* lds, int rb, int n,
                                           int lane, int half) {
  const int j = (lane & 15) >> 2;
  const int dq = n * 8 + ((lane & 16) >> 2) + (lane & 3);
  const int r0 = rb + half * 8 + j;
  const int r1 = r0 + 4;
  const unsigned base = (unsigned)(uintptr_t)lds;
  const unsigned a0 = base + r0 * 256 + ((dq * 8) ^ ((r0 & 7) << 4));
  const unsigned a1 = base + r1 * 256 + ((dq * 8) ^ ((r1 & 7) << 4));
  u32x2_t w0, w1;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=v"(w0), "=v"(w1)
      : "v"(a0), "v"(a1));
  u32x4_t frag = u32x4_t{w0.x, w0.y, w1.x, w1.y};
  return *reinterpret_cast<bf16x8*>(&frag);
}

// B-operand fragments [16 rows, 32 cols] assembled from C-form per-lane
// values v16 (16 f32, C rows (r&3)+8*(r>>2)+4*half, col = lane-own): two
// k-slices of 16 rows each. See attention_v3.hip P^T derivation.
__device__ __forceinline__ void pairswap_frags(const float v16[16], int half,
                                               bf16x8 out[2]) {
#pragma unroll
  for (int ks2 = 0; ks2 < 2; ++ks2) {
    const int r0 = ks2 * 8;
    unsigned w0 = pk_bf16(v16[r0 + 0], v16[r0 + 1]);
    unsigned w1 = pk_bf16(v16[r0 + 2], v16[r0 + 3]);
    unsigned w2 = pk_bf16(v16[r0 + 4], v16[r0 + 5]);
    unsigned w3 = pk_bf16(v16[r0 + 6], v16[r0 + 7]);
    u32x2_t s0 = __builtin_amdgcn_permlane32_swap(w0, w2, false, false);
    u32x2_t s1 = __builtin_amdgcn_permlane32_swap(w1, w3, false, false);
    u32x4_t frag = u32x4_t{s0.x, s1.x, s0.y, s1.y};
    out[ks2] = *reinterpret_cast<bf16x8*>(&frag);
  }
  (void)half;  // order is half-independent (see fwd derivation)
}

// ---------------------------------------------------------------------------
// dQ kernel
// ---------------------------------------------------------------------------

__global__ __launch_bounds__(512) void fa_bwd_dq_v3_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ dvec,
    short* __restrict__ dq, int B, int H, int HKV, int S, float scale,
    long long qs_b, long long qs_h, long long qs_s,
    long long ks_b, long long ks_h, long long ks_s,
    long long vs_b, long long vs_h, long long vs_s,
    long long ds_b, long long ds_h, long long ds_s) {
  __shared__ char k_lds[2][FB_T * FB_D * 2];  // 2 x 16 KB, row-major swz
  __shared__ char v_lds[2][FB_T * FB_D * 2];  // 2 x 16 KB, row-major swz

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int half = lane >> 5;
  const int qcol = lane & 31;

  const int qt = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int g = h / (H / HKV);

  const short* q_blk = q + (long long)b * qs_b + (long long)h * qs_h +
                       (long long)(qt * FB_QB) * qs_s;
  const short* do_blk = dout + (long long)b * ds_b + (long long)h * ds_h +
                        (long long)(qt * FB_QB) * ds_s;
  const short* k_head = k + (long long)b * ks_b + (long long)g * ks_h;
  const short* v_head = v + (long long)b * vs_b + (long long)g * vs_h;

  const int q_row_w = wave * 32 + qcol;
  const int q_glob = qt * FB_QB + q_row_w;

  // persistent Q and dO rows as B-operand fragments
  bf16x8 bq[8], bdo[8];
#pragma unroll
  for (int ks = 0; ks < 8; ++ks) {
    bq[ks] = *reinterpret_cast<const bf16x8*>(
        q_blk + (long long)q_row_w * qs_s + ks * 16 + half * 8);
    bdo[ks] = *reinterpret_cast<const bf16x8*>(
        do_blk + (long long)q_row_w * ds_s + ks * 16 + half * 8);
  }
  const long long lrow = ((long long)b * H + h) * S + q_glob;
  const float L = lse[lrow];
  const float Dv = dvec[lrow];

  f32x16_t acc_dq[4];
#pragma unroll
  for (int n = 0; n < 4; ++n)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc_dq[n][r] = 0.f;

  // staging (T14): 2 x b128 of K and V per thread
  bf16x8 st_k[2], st_v[2];
  const int st_row0 = (tid * 16) / FB_D;
  const int st_col0 = (tid * 16) % FB_D;
  auto issue_loads = [&](int kt) {
    const short* ksrc = k_head + (long long)(kt * FB_T) * ks_s;
    const short* vsrc = v_head + (long long)(kt * FB_T) * vs_s;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      st_k[c] = *reinterpret_cast<const bf16x8*>(
          ksrc + (long long)st_row0 * ks_s + st_col0 + c * 8);
      st_v[c] = *reinterpret_cast<const bf16x8*>(
          vsrc + (long long)st_row0 * vs_s + st_col0 + c * 8);
    }
  };
  auto write_lds = [&](int buf) {
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int col = st_col0 + c * 8;
      *reinterpret_cast<bf16x8*>(k_lds[buf] + st_row0 * 256 +
                                 swzb(st_row0, col * 2)) = st_k[c];
      *reinterpret_cast<bf16x8*>(v_lds[buf] + st_row0 * 256 +
                                 swzb(st_row0, col * 2)) = st_v[c];
    }
  };

  const int n_tiles = (qt + 1) * (FB_QB / FB_T);
  const int wave_q_max = qt * FB_QB + wave * 32 + 31;

  issue_loads(0);
  write_lds(0);
  __syncthreads();

  for (int kt = 0; kt < n_tiles; ++kt) {
    const int cur = kt & 1;
    const char* k_cur = k_lds[cur];
    const char* v_cur = v_lds[cur];
    if (kt + 1 < n_tiles) issue_loads(kt + 1);

    const int kv0 = kt * FB_T;
    if (kv0 <= wave_q_max) {
      // per kv-subtile of 32: S^T -> P^T -> dP^T -> dS^T frags, one
      // accumulator pair live at a time (register pressure: the 2-subtile
      // batched version spilled 26 VGPRs)
      bf16x8 bds[4];
#pragma unroll
      for (int n = 0; n < 2; ++n) {
        f32x16_t acc_s;
#pragma unroll
        for (int r = 0; r < 16; ++r) acc_s[r] = 0.f;
#pragma unroll
        for (int ks = 0; ks < 8; ++ks) {
          const int row = n * 32 + qcol;
          bf16x8 ak = *reinterpret_cast<const bf16x8*>(
              k_cur + row * 256 + swzb(row, (ks * 16 + half * 8) * 2));
          acc_s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ak, bq[ks], acc_s,
                                                          0, 0, 0);
        }
        // P^T = exp(S^T*scale - L), causal-zeroed
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kv_glob = kv0 + n * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          acc_s[r] =
              (kv_glob > q_glob) ? 0.f : __expf(acc_s[r] * scale - L);
        }
        // dP^T = V dO^T
        f32x16_t acc_dp;
#pragma unroll
        for (int r = 0; r < 16; ++r) acc_dp[r] = 0.f;
#pragma unroll
        for (int ks = 0; ks < 8; ++ks) {
          const int row = n * 32 + qcol;
          bf16x8 av = *reinterpret_cast<const bf16x8*>(
              v_cur + row * 256 + swzb(row, (ks * 16 + half * 8) * 2));
          acc_dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(av, bdo[ks],
                                                           acc_dp, 0, 0, 0);
        }
        // dS^T = P^T .* (dP^T - Dv) * scale -> B frags
        float dsv[16];
#pragma unroll
        for (int r = 0; r < 16; ++r)
          dsv[r] = acc_s[r] * (acc_dp[r] - Dv) * scale;
        pairswap_frags(dsv, half, &bds[n * 2]);
      }
      // ---- dQ^T += K^T dS^T ----
#pragma unroll
      for (int n = 0; n < 4; ++n) {
#pragma unroll
        for (int ks = 0; ks < 4; ++ks) {
          bf16x8 akT = trT_frag(k_cur, ks * 16, n, lane, half);
          acc_dq[n] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              akT, bds[ks], acc_dq[n], 0, 0, 0);
        }
      }
    }

    __syncthreads();
    if (kt + 1 < n_tiles) {
      write_lds(cur ^ 1);
      __syncthreads();
    }
  }

  // ---- epilogue: dQ^T -> LDS bounce -> coalesced stores ----
  // dq uses the SAME strides as q (the binding stages q into dq's layout,
  // v1 convention)
  short* dq_blk = dq + (long long)b * qs_b + (long long)h * qs_h +
                  (long long)(qt * FB_QB) * qs_s;
  char* slab = (char*)k_lds + (wave & 3) * (32 * FB_D * 2);
  for (int round = 0; round < 2; ++round) {
    __syncthreads();
    if ((wave >> 2) == round) {
#pragma unroll
      for (int n = 0; n < 4; ++n) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int d = n * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          *reinterpret_cast<__bf16*>(slab + qcol * 256 + swzb(qcol, d * 2)) =
              (__bf16)acc_dq[n][r];
        }
      }
    }
    __syncthreads();
    if ((wave >> 2) == round) {
#pragma unroll
      for (int pass = 0; pass < 4; ++pass) {
        const int elem0 = (pass * 64 + lane) * 16;
        const int r2 = elem0 / FB_D;
        const int c2 = elem0 % FB_D;
        bf16x8 v0 = *reinterpret_cast<const bf16x8*>(
            slab + r2 * 256 + swzb(r2, c2 * 2));
        bf16x8 v1 = *reinterpret_cast<const bf16x8*>(
            slab + r2 * 256 + swzb(r2, (c2 + 8) * 2));
        short* dst = dq_blk + (long long)(wave * 32 + r2) * qs_s + c2;
        *reinterpret_cast<bf16x8*>(dst) = v0;
        *reinterpret_cast<bf16x8*>(dst + 8) = v1;
      }
    }
  }
}

// ---------------------------------------------------------------------------
// dKV kernel (two passes over the q range: dV then dK — batching both sets
// of 64-VGPR accumulators spilled; the S recompute is cheaper than spills)
// ---------------------------------------------------------------------------

// shared prologue macro for the two kv-major kernels
#define FB_DKV_PROLOGUE()                                                    \
  __shared__ char q_lds[FB_T * FB_D * 2];                                    \
  __shared__ char do_lds[FB_T * FB_D * 2];                                   \
  __shared__ float l_lds[FB_T];                                              \
  __shared__ float d_lds[FB_T];                                              \
  const int tid = threadIdx.x;                                               \
  const int wave = tid >> 6;                                                 \
  const int lane = tid & 63;                                                 \
  const int half = lane >> 5;                                                \
  const int kcol = lane & 31;                                                \
  const int kvb0 = blockIdx.x * FB_QB;                                       \
  const int chunk = blockIdx.y;                                              \
  const int CH = gridDim.y;                                                  \
  const int bg = blockIdx.z;                                                 \
  const int b = bg / HKV;                                                    \
  const int g = bg % HKV;                                                    \
  const int rep = H / HKV;                                                   \
  const int kv_own = kvb0 + wave * 32 + kcol;                                \
  const short* k_head = k + (long long)b * ks_b + (long long)g * ks_h;       \
  bf16x8 bk[8];                                                              \
  _Pragma("unroll") for (int ks = 0; ks < 8; ++ks) bk[ks] =                  \
      *reinterpret_cast<const bf16x8*>(k_head + (long long)kv_own * ks_s +   \
                                       ks * 16 + half * 8);                  \
  const int qt_min = kvb0 / FB_T;                                            \
  const int n_qt = S / FB_T - qt_min;                                        \
  const int n_work = rep * n_qt;                                             \
  const int wave_kv_min = kvb0 + wave * 32;                                  \
  auto stage_qdo = [&](int h, int qt) {                                      \
    const short* qsrc = q + (long long)b * qs_b + (long long)h * qs_h +      \
                        (long long)(qt * FB_T) * qs_s;                       \
    const short* dsrc = dout + (long long)b * ds_b + (long long)h * ds_h +   \
                        (long long)(qt * FB_T) * ds_s;                       \
    _Pragma("unroll") for (int c = 0; c < 2; ++c) {                          \
      const int linear = (tid * 16) + c * 8;                                 \
      const int row = linear / FB_D;                                         \
      const int col = linear % FB_D;                                         \
      *reinterpret_cast<bf16x8*>(q_lds + row * 256 + swzb(row, col * 2)) =   \
          *reinterpret_cast<const bf16x8*>(qsrc + (long long)row * qs_s +    \
                                           col);                             \
      *reinterpret_cast<bf16x8*>(do_lds + row * 256 + swzb(row, col * 2)) =  \
          *reinterpret_cast<const bf16x8*>(dsrc + (long long)row * ds_s +    \
                                           col);                             \
    }                                                                        \
    if (tid < FB_T) {                                                        \
      const long long lrow = ((long long)b * H + h) * S + qt * FB_T + tid;   \
      l_lds[tid] = lse[lrow];                                                \
      d_lds[tid] = dvec[lrow];                                               \
    }                                                                        \
  }

__global__ __launch_bounds__(512) void fa_bwd_dv_v3_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ dvec,
    float* __restrict__ dk32, float* __restrict__ dv32, int B, int H,
    int HKV, int S, float scale,
    long long qs_b, long long qs_h, long long qs_s,
    long long ks_b, long long ks_h, long long ks_s,
    long long vs_b, long long vs_h, long long vs_s,
    long long ds_b, long long ds_h, long long ds_s) {
  FB_DKV_PROLOGUE();

  // ---------------- pass 1: dV^T = dO^T P ----------------
  {
    f32x16_t acc_dv[4];
#pragma unroll
    for (int n = 0; n < 4; ++n)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc_dv[n][r] = 0.f;

    for (int w = chunk; w < n_work; w += CH) {
      const int h = g * rep + w / n_qt;
      const int qt = qt_min + w % n_qt;
      __syncthreads();
      stage_qdo(h, qt);
      __syncthreads();
      if (qt * FB_T + FB_T - 1 < wave_kv_min) continue;  // fully masked
#pragma unroll
      for (int st = 0; st < 2; ++st) {  // q subtiles of 32
        const int q0 = qt * FB_T + st * 32;
        if (q0 + 31 < wave_kv_min) continue;
        // S = Q K^T (C col = kv own, rows = q)
        f32x16_t acc_s;
#pragma unroll
        for (int r = 0; r < 16; ++r) acc_s[r] = 0.f;
#pragma unroll
        for (int ks = 0; ks < 8; ++ks) {
          const int row = st * 32 + kcol;  // q row index == lane&31 pattern
          bf16x8 aq = *reinterpret_cast<const bf16x8*>(
              q_lds + (st * 32 + kcol) * 256 +
              swzb(st * 32 + kcol, (ks * 16 + half * 8) * 2));
          (void)row;
          acc_s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(aq, bk[ks], acc_s,
                                                          0, 0, 0);
        }
        // P = exp(S*scale - L[q]), causal mask kv > q
        float pv[16];
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qrow = st * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          const int q_glob = qt * FB_T + qrow;
          pv[r] = (kv_own > q_glob)
                      ? 0.f
                      : __expf(acc_s[r] * scale - l_lds[qrow]);
        }
        bf16x8 bp[2];
        pairswap_frags(pv, half, bp);
        // dV^T += dO^T P
#pragma unroll
        for (int n = 0; n < 4; ++n) {
#pragma unroll
          for (int ks2 = 0; ks2 < 2; ++ks2) {
            bf16x8 adoT =
                trT_frag(do_lds, st * 32 + ks2 * 16, n, lane, half);
            acc_dv[n] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                adoT, bp[ks2], acc_dv[n], 0, 0, 0);
          }
        }
      }
    }
    // flush dV (fp32 atomics: chunks and — at CH=1 — nothing else touches
    // these elements, but atomics keep the chunked path correct)
    float* dv_base = dv32 + ((long long)bg * S + kv_own) * FB_D;
#pragma unroll
    for (int n = 0; n < 4; ++n)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = n * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
        atomicAdd(dv_base + d, acc_dv[n][r]);
      }
  }

}

__global__ __launch_bounds__(512) void fa_bwd_dk_v3_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ dvec,
    float* __restrict__ dk32, float* __restrict__ dv32, int B, int H,
    int HKV, int S, float scale,
    long long qs_b, long long qs_h, long long qs_s,
    long long ks_b, long long ks_h, long long ks_s,
    long long vs_b, long long vs_h, long long vs_s,
    long long ds_b, long long ds_h, long long ds_s) {
  FB_DKV_PROLOGUE();
  const short* v_head = v + (long long)b * vs_b + (long long)g * vs_h;
  // persistent V row (B-operand for dP = dO V^T)
  bf16x8 bv[8];
#pragma unroll
  for (int ks = 0; ks < 8; ++ks)
    bv[ks] = *reinterpret_cast<const bf16x8*>(
        v_head + (long long)kv_own * vs_s + ks * 16 + half * 8);

  // ---------------- pass 2: dK^T = Q^T dS ----------------
  {
    f32x16_t acc_dk[4];
#pragma unroll
    for (int n = 0; n < 4; ++n)
#pragma unroll
      for (int r = 0; r < 16; ++r) acc_dk[n][r] = 0.f;

    for (int w = chunk; w < n_work; w += CH) {
      const int h = g * rep + w / n_qt;
      const int qt = qt_min + w % n_qt;
      __syncthreads();
      stage_qdo(h, qt);
      __syncthreads();
      if (qt * FB_T + FB_T - 1 < wave_kv_min) continue;
#pragma unroll
      for (int st = 0; st < 2; ++st) {
        const int q0 = qt * FB_T + st * 32;
        if (q0 + 31 < wave_kv_min) continue;
        // S = Q K^T and dP = dO V^T in ONE interleaved k-loop (separate
        // 8-deep unrolled loops kept too many LDS-read temporaries live:
        // 62 VGPR spills)
        f32x16_t acc_s, acc_dp;
#pragma unroll
        for (int r = 0; r < 16; ++r) acc_s[r] = acc_dp[r] = 0.f;
#pragma unroll
        for (int ks = 0; ks < 8; ++ks) {
          const int row = st * 32 + kcol;
          const int bc = (ks * 16 + half * 8) * 2;
          bf16x8 aq = *reinterpret_cast<const bf16x8*>(
              q_lds + row * 256 + swzb(row, bc));
          acc_s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(aq, bk[ks], acc_s,
                                                          0, 0, 0);
          bf16x8 ado = *reinterpret_cast<const bf16x8*>(
              do_lds + row * 256 + swzb(row, bc));
          acc_dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ado, bv[ks],
                                                           acc_dp, 0, 0, 0);
        }
        // dS = P .* (dP - Dvec[q]) * scale
        float dsv[16];
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int qrow = st * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          const int q_glob = qt * FB_T + qrow;
          const float p = (kv_own > q_glob)
                              ? 0.f
                              : __expf(acc_s[r] * scale - l_lds[qrow]);
          dsv[r] = p * (acc_dp[r] - d_lds[qrow]) * scale;
        }
        bf16x8 bds[2];
        pairswap_frags(dsv, half, bds);
        // dK^T += Q^T dS
#pragma unroll
        for (int n = 0; n < 4; ++n) {
#pragma unroll
          for (int ks2 = 0; ks2 < 2; ++ks2) {
            bf16x8 aqT = trT_frag(q_lds, st * 32 + ks2 * 16, n, lane, half);
            acc_dk[n] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                aqT, bds[ks2], acc_dk[n], 0, 0, 0);
          }
        }
      }
    }
    float* dk_base = dk32 + ((long long)bg * S + kv_own) * FB_D;
#pragma unroll
    for (int n = 0; n < 4; ++n)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int d = n * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
        atomicAdd(dk_base + d, acc_dk[n][r]);
      }
  }
}

extern "C" void fa_bwd_dkv_v3_launch(const void* q, const void* k,
                                     const void* v, const void* dout,
                                     const void* lse, const void* dvec,
                                     void* dk32, void* dv32, int B, int H,
                                     int HKV, int S, float scale,
                                     const long long* strides,
                                     hipStream_t stream) {
  const int kv_blocks = S / FB_QB;
  // chunk (head, q-tile) work across gridDim.y until the launch fills the
  // 256-CU chip (each block is CU-wide: 512 threads, 1 block/CU)
  int ch = 1;
  while (kv_blocks * ch * B * HKV < 512 && ch < 16) ch <<= 1;
  dim3 grid(kv_blocks, ch, B * HKV);
  hipLaunchKernelGGL(fa_bwd_dv_v3_kernel, grid, dim3(512), 0, stream,
                     (const short*)q, (const short*)k, (const short*)v,
                     (const short*)dout, (const float*)lse,
                     (const float*)dvec, (float*)dk32, (float*)dv32, B, H,
                     HKV, S, scale, strides[0], strides[1], strides[2],
                     strides[3], strides[4], strides[5], strides[6],
                     strides[7], strides[8], strides[9], strides[10],
                     strides[11]);
  hipLaunchKernelGGL(fa_bwd_dk_v3_kernel, grid, dim3(512), 0, stream,
                     (const short*)q, (const short*)k, (const short*)v,
                     (const short*)dout, (const float*)lse,
                     (const float*)dvec, (float*)dk32, (float*)dv32, B, H,
                     HKV, S, scale, strides[0], strides[1], strides[2],
                     strides[3], strides[4], strides[5], strides[6],
                     strides[7], strides[8], strides[9], strides[10],
                     strides[11]);
}

extern "C" void fa_bwd_dq_v3_launch(const void* q, const void* k,
                                    const void* v, const void* dout,
                                    const void* lse, const void* dvec,
                                    void* dq, int B, int H, int HKV, int S,
                                    float scale, const long long* strides,
                                    hipStream_t stream) {
  dim3 grid(S / FB_QB, H, B);
  hipLaunchKernelGGL(fa_bwd_dq_v3_kernel, grid, dim3(512), 0, stream,
                     (const short*)q, (const short*)k, (const short*)v,
                     (const short*)dout, (const float*)lse,
                     (const float*)dvec, (short*)dq, B, H, HKV, S, scale,
                     strides[0], strides[1], strides[2], strides[3],
                     strides[4], strides[5], strides[6], strides[7],
                     strides[8], strides[9], strides[10], strides[11]);
}
