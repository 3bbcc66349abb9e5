// Flash attention (forward) for MI355X: MFMA 16x16x32 bf16, online softmax,
// causal, GQA, head_dim=128.
//
// The reference has no attention kernel of its own (SURVEY.md §2.3 — it only
// OBSERVES FlashAttention launches in xpu_timer); this is part of the
// MI355X-native hot path. Design per the CDNA4 guide (§B fused attention):
// never materialize the S x S score matrix; per Q-tile iterate K/V-tiles with
// running (m, l) rescaling.
//
// Geometry: block = 256 threads = 4 waves; one block owns one
// (batch, q-head, 64-row q-tile); wave w owns q rows [16w, 16w+16).
// Per KV tile (64 rows):
//   K staged in LDS row-major [64][128] with the guide's XOR swizzle
//   (byte ^= (row&7)<<4) so B-fragment ds_read_b128 is conflict-light;
//   V staged TRANSPOSED [128][64] (so the PV B-fragment reads are contiguous)
//   with the same swizzle;
//   S_band[16,64] = Q_band @ K^T  (16 MFMA per wave),
//   online-softmax update, P staged through LDS (swizzled) to re-layout for
//   the A operand, O_band[16,128] += P_band @ V (16 MFMA per wave).
// Saves per-row logsumexp L for the backward.
//
// MFMA operand layouts (hardware-verified by tests/test_mfma_gpu.py):
//   A[16,32]: lane l -> row l%16, k = (l/16)*8 + [0,8)
//   B[32,16]: lane l -> k = (l/16)*8 + [0,8), col l%16
//   C[16,16]: lane l, reg r -> row (l>>4)*4 + r, col l&15
#include "kern_common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define FA_D 128
#define FA_QT 64  // q rows per block
#define FA_KT 64  // kv rows per tile

// XOR swizzle on a byte offset within a 128-byte LDS row group (guide G4)
__device__ __forceinline__ int swz(int row, int byte_col) {
  return (row * 0 + byte_col) ^ ((row & 7) << 4);
}

// wave-local reduction across the 16 lanes that share a C-row quarter
__device__ __forceinline__ float qmax16(float v) {
#pragma unroll
  for (int s = 8; s > 0; s >>= 1) v = fmaxf(v, __shfl_xor(v, s, 64));
  return v;
}
__device__ __forceinline__ float qsum16(float v) {
#pragma unroll
  for (int s = 8; s > 0; s >>= 1) v += __shfl_xor(v, s, 64);
  return v;
}

__device__ __forceinline__ bf16x8 ld_frag_b128(const char* lds_base, int row,
                                               int byte_col) {
  const int off = row * 256 + swz(row, byte_col);
  return *reinterpret_cast<const bf16x8*>(lds_base + off);
}

// All tensors are logically [B, H, S, D] with EXPLICIT strides (element
// units, D contiguous) — callers pass permuted views, so [B,S,H,D] storage
// needs no transpose copies.
__global__ __launch_bounds__(256) void flash_attn_fwd_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ out,
    float* __restrict__ lse, int B, int H, int HKV, int S, float scale,
    long long qs_b, long long qs_h, long long qs_s,
    long long ks_b, long long ks_h, long long ks_s,
    long long vs_b, long long vs_h, long long vs_s,
    long long os_b, long long os_h, long long os_s) {
  // LDS: K [64][128] bf16 (rows padded to 256B as-is), V^T [128][64] bf16,
  // P [4 waves][16][64] bf16 — all swizzled.
  // double-buffered K/V so the NEXT tile's HBM loads overlap this tile's
  // MFMA (guide §5.5 T3 minimum 2-phase: stage-next before compute, one
  // barrier pair per tile). 72 KB total still fits 2 blocks/CU.
  __shared__ char k_lds[2][FA_KT * FA_D * 2];     // 2 x 16 KB
  __shared__ char vt_lds[2][FA_D * FA_KT * 2];    // 2 x 16 KB
  __shared__ char p_lds[4 * 16 * FA_KT * 2];      // 8 KB

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int quarter = lane >> 4;  // 0..3
  const int sub = lane & 15;      // 0..15

  const int qt = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int g = h / (H / HKV);  // GQA kv head

  const short* q_blk = q + (long long)b * qs_b + (long long)h * qs_h +
                       (long long)(qt * FA_QT) * qs_s;
  const short* k_head = k + (long long)b * ks_b + (long long)g * ks_h;
  const short* v_head = v + (long long)b * vs_b + (long long)g * vs_h;

  // ---- load this wave's Q band [16,128] into A fragments (persistent) ----
  // frag ks (k-slice of 32): lane holds Q[qrow0 + sub][ks*32 + quarter*8 + i]
  const int qrow_w = wave * 16;
  bf16x8 aq[4];
#pragma unroll
  for (int ks = 0; ks < 4; ++ks) {
    const short* src =
        q_blk + (long long)(qrow_w + sub) * qs_s + ks * 32 + quarter * 8;
    aq[ks] = *reinterpret_cast<const bf16x8*>(src);
  }

  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -1e30f;
    l_run[r] = 0.f;
  }
  f32x4_t acc_o[8];
#pragma unroll
  for (int n = 0; n < 8; ++n) acc_o[n] = {0.f, 0.f, 0.f, 0.f};

  char* p_wave = p_lds + wave * 16 * FA_KT * 2;

  // ---- stage one K/V tile into buffer `buf` ----
  auto stage = [&](int kt, int buf) {
    const short* ksrc = k_head + (long long)(kt * FA_KT) * ks_s;
    const short* vsrc = v_head + (long long)(kt * FA_KT) * vs_s;
    char* kb = k_lds[buf];
    char* vb = vt_lds[buf];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int linear = (tid * 4 + c) * 8;  // 8-elem chunk start
      const int row = linear / FA_D;
      const int col = linear % FA_D;
      bf16x8 kv8 =
          *reinterpret_cast<const bf16x8*>(ksrc + (long long)row * ks_s + col);
      *reinterpret_cast<bf16x8*>(kb + row * 256 + swz(row, col * 2)) = kv8;
      bf16x8 vv8 =
          *reinterpret_cast<const bf16x8*>(vsrc + (long long)row * vs_s + col);
#pragma unroll
      for (int i = 0; i < 8; ++i) {
        const int trow = col + i;  // head-dim index
        const int tcol = row;      // kv position
        *reinterpret_cast<__bf16*>(vb + trow * 128 + swz(trow, tcol * 2)) =
            vv8[i];
      }
    }
  };

  stage(0, 0);
  __syncthreads();

  for (int kt = 0; kt <= qt; ++kt) {
    const int cur = kt & 1;
    char* k_cur = k_lds[cur];
    char* vt_cur = vt_lds[cur];
    // prefetch the NEXT tile into the other buffer while computing this one
    if (kt < qt) stage(kt + 1, cur ^ 1);

    // ---- S_band = Q_band @ K^T ----
    f32x4_t acc_s[4];
#pragma unroll
    for (int n = 0; n < 4; ++n) acc_s[n] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        // B[k][col] = K[n*16 + sub][ks*32 + quarter*8 + i] (K^T fragment)
        bf16x8 bk;
        {
          const int row = n * 16 + sub;
          const int byte_col = (ks * 32 + quarter * 8) * 2;
          bk = *reinterpret_cast<const bf16x8*>(
              k_cur + row * 256 + swz(row, byte_col));
        }
        acc_s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[ks], bk,
                                                           acc_s[n], 0, 0, 0);
      }
    }

    // ---- causal mask + online softmax ----
    const int row_glob = qt * FA_QT + qrow_w + quarter * 4;  // + r
    float pmax[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) pmax[r] = -1e30f;
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const int col_glob = kt * FA_KT + n * 16 + sub;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float s = acc_s[n][r] * scale;
        if (col_glob > row_glob + r) s = -1e30f;
        acc_s[n][r] = s;
        pmax[r] = fmaxf(pmax[r], s);
      }
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float m_new = fmaxf(m_run[r], qmax16(pmax[r]));
      alpha[r] = __expf(m_run[r] - m_new);
      m_run[r] = m_new;
    }
    // P = exp(s - m), row sums, stage P^T-layout into LDS
    float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = __expf(acc_s[n][r] - m_run[r]);
        psum[r] += p;
        // P row = quarter*4 + r, col = n*16 + sub
        const int prow = quarter * 4 + r;
        const int pcol = n * 16 + sub;
        *reinterpret_cast<__bf16*>(p_wave + prow * 128 + swz(prow, pcol * 2)) =
            (__bf16)p;
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      l_run[r] = l_run[r] * alpha[r] + qsum16(psum[r]);
#pragma unroll
      for (int n = 0; n < 8; ++n) {
        // rescale only the rows this (quarter, r) owns — all acc rows share
        // the lane's quarter, so alpha[r] applies to element r of every frag
        acc_o[n][r] *= alpha[r];
      }
    }
    __syncthreads();  // P staged + next-tile K/V writes landed

    // ---- O_band += P_band @ V ----
#pragma unroll
    for (int n = 0; n < 8; ++n) {
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        // A fragment: P[row sub][ks*32 + quarter*8 + i]
        bf16x8 ap;
        {
          const int row = sub;
          const int byte_col = (ks * 32 + quarter * 8) * 2;
          ap = *reinterpret_cast<const bf16x8*>(
              p_wave + row * 128 + swz(row, byte_col));
        }
        // B fragment: V[k][n*16+sub] = VT[n*16+sub][k], k = ks*32+quarter*8+i
        bf16x8 bv;
        {
          const int row = n * 16 + sub;
          const int byte_col = (ks * 32 + quarter * 8) * 2;
          bv = *reinterpret_cast<const bf16x8*>(
              vt_cur + row * 128 + swz(row, byte_col));
        }
        acc_o[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, bv, acc_o[n],
                                                           0, 0, 0);
      }
    }
    __syncthreads();  // all reads of buf[cur] done before kt+2 overwrites it
  }

  // ---- epilogue: normalize, store O and logsumexp ----
  short* o_blk = out + (long long)b * os_b + (long long)h * os_h +
                 (long long)(qt * FA_QT + qrow_w) * os_s;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const float inv = l_run[r] > 0.f ? 1.f / l_run[r] : 0.f;
    const int orow = quarter * 4 + r;
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      o_blk[(long long)orow * os_s + n * 16 + sub] = f2bf(acc_o[n][r] * inv);
    }
    if (sub == 0) {
      const long long lrow =
          ((long long)b * H + h) * S + qt * FA_QT + qrow_w + orow;
      lse[lrow] = m_run[r] + __logf(fmaxf(l_run[r], 1e-30f));
    }
  }
}

extern "C" void flash_attn_fwd_launch(const void* q, const void* k,
                                      const void* v, void* out, void* lse,
                                      int B, int H, int HKV, int S,
                                      float scale, const long long* strides,
                                      hipStream_t stream) {
  dim3 grid(S / FA_QT, H, B);
  hipLaunchKernelGGL(flash_attn_fwd_kernel, grid, dim3(256), 0, stream,
                     (const short*)q, (const short*)k, (const short*)v,
                     (short*)out, (float*)lse, B, H, HKV, S, scale,
                     strides[0], strides[1], strides[2], strides[3],
                     strides[4], strides[5], strides[6], strides[7],
                     strides[8], strides[9], strides[10], strides[11]);
}

// ============================================================================
// V2: same schedule as flash_attn_fwd_kernel but V staged in a 4x4-subtiled
// layout consumed with gfx950's ds_read_b64_tr_b16 hardware transpose read.
//
// Probe-verified semantics (tests/test_mfma_gpu.py::test_tr_b16_probe, run on
// MI355X): a lane reads column (A/2 & 3) of the row-major 4x4 bf16 tile at
// byte (A & ~0x18) — 4 bf16 from bytes (A&~0x18) + {0,8,16,24}. So V is
// stored as tiles [cb = col/4][kb = k/4][k&3][col&3] (32 B each, kb-major
// within a cb column: consecutive k-tiles 32 B apart), and the PV B-fragment
// (lane wants V[ks*32 + quarter*8 + i][n*16 + sub]) is exactly two tr reads
// at kb = ks*8 + quarter*2 and kb+1.
//
// Staging writes become two b64 stores per 8-element chunk (vs 8 scalar
// stores in v1 — the measured 35%-of-cycles bank-conflict source,
// profiles/r01e). XOR key ((cb>>3)&3)<<5 spreads the 4 lanes that share
// (k&15) across kb-groups: quarter-wave write coverage is conflict-free.
// The key is uniform across a tr read's 4-element gather (it depends only
// on cb), so reads apply the same key to their computed address; the
// offset:N immediate is NOT used across tiles because + does not commute
// with ^ on the keyed bits.

__device__ __forceinline__ int v4_addr(int k, int c) {
  // byte address of element (k row, c col) in the subtiled V buffer
  const int base = ((c >> 2) * 512 + (k >> 2) * 32) ^ (((c >> 5) & 3) << 5);
  return base + (k & 3) * 8 + (c & 3) * 2;
}

typedef __attribute__((ext_vector_type(2))) unsigned int u32x2_t;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4_t;

__global__ __launch_bounds__(256) void flash_attn_fwd_v2_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ out,
    float* __restrict__ lse, int B, int H, int HKV, int S, float scale,
    long long qs_b, long long qs_h, long long qs_s,
    long long ks_b, long long ks_h, long long ks_s,
    long long vs_b, long long vs_h, long long vs_s,
    long long os_b, long long os_h, long long os_s) {
  __shared__ char k_lds[2][FA_KT * FA_D * 2];     // 2 x 16 KB
  __shared__ char v4_lds[2][FA_KT * FA_D * 2];    // 2 x 16 KB (subtiled)
  __shared__ char p_lds[4 * 16 * FA_KT * 2];      // 8 KB

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int quarter = lane >> 4;
  const int sub = lane & 15;

  const int qt = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int g = h / (H / HKV);

  const short* q_blk = q + (long long)b * qs_b + (long long)h * qs_h +
                       (long long)(qt * FA_QT) * qs_s;
  const short* k_head = k + (long long)b * ks_b + (long long)g * ks_h;
  const short* v_head = v + (long long)b * vs_b + (long long)g * vs_h;

  const int qrow_w = wave * 16;
  bf16x8 aq[4];
#pragma unroll
  for (int ks = 0; ks < 4; ++ks) {
    const short* src =
        q_blk + (long long)(qrow_w + sub) * qs_s + ks * 32 + quarter * 8;
    aq[ks] = *reinterpret_cast<const bf16x8*>(src);
  }

  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -1e30f;
    l_run[r] = 0.f;
  }
  f32x4_t acc_o[8];
#pragma unroll
  for (int n = 0; n < 8; ++n) acc_o[n] = {0.f, 0.f, 0.f, 0.f};

  char* p_wave = p_lds + wave * 16 * FA_KT * 2;

  auto stage = [&](int kt, int buf) {
    const short* ksrc = k_head + (long long)(kt * FA_KT) * ks_s;
    const short* vsrc = v_head + (long long)(kt * FA_KT) * vs_s;
    char* kb = k_lds[buf];
    char* vb = v4_lds[buf];
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int linear = (tid * 4 + c) * 8;
      const int row = linear / FA_D;
      const int col = linear % FA_D;
      bf16x8 kv8 =
          *reinterpret_cast<const bf16x8*>(ksrc + (long long)row * ks_s + col);
      *reinterpret_cast<bf16x8*>(kb + row * 256 + swz(row, col * 2)) = kv8;
      bf16x8 vv8 =
          *reinterpret_cast<const bf16x8*>(vsrc + (long long)row * vs_s + col);
      const u32x4_t vbits = *reinterpret_cast<const u32x4_t*>(&vv8);
      // elems 0..3 -> tile column col, elems 4..7 -> tile column col+4
      *reinterpret_cast<u32x2_t*>(vb + v4_addr(row, col)) =
          u32x2_t{vbits.x, vbits.y};
      *reinterpret_cast<u32x2_t*>(vb + v4_addr(row, col + 4)) =
          u32x2_t{vbits.z, vbits.w};
    }
  };

  stage(0, 0);
  __syncthreads();

  for (int kt = 0; kt <= qt; ++kt) {
    const int cur = kt & 1;
    char* k_cur = k_lds[cur];
    char* v4_cur = v4_lds[cur];
    if (kt < qt) stage(kt + 1, cur ^ 1);

    // ---- S_band = Q_band @ K^T (unchanged from v1) ----
    f32x4_t acc_s[4];
#pragma unroll
    for (int n = 0; n < 4; ++n) acc_s[n] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8 bk;
        {
          const int row = n * 16 + sub;
          const int byte_col = (ks * 32 + quarter * 8) * 2;
          bk = *reinterpret_cast<const bf16x8*>(
              k_cur + row * 256 + swz(row, byte_col));
        }
        acc_s[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(aq[ks], bk,
                                                           acc_s[n], 0, 0, 0);
      }
    }

    // ---- causal mask + online softmax (unchanged) ----
    const int row_glob = qt * FA_QT + qrow_w + quarter * 4;
    float pmax[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) pmax[r] = -1e30f;
#pragma unroll
    for (int n = 0; n < 4; ++n) {
      const int col_glob = kt * FA_KT + n * 16 + sub;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float s = acc_s[n][r] * scale;
        if (col_glob > row_glob + r) s = -1e30f;
        acc_s[n][r] = s;
        pmax[r] = fmaxf(pmax[r], s);
      }
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const float m_new = fmaxf(m_run[r], qmax16(pmax[r]));
      alpha[r] = __expf(m_run[r] - m_new);
      m_run[r] = m_new;
    }
    float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int n = 0; n < 4; ++n) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const float p = __expf(acc_s[n][r] - m_run[r]);
        psum[r] += p;
        const int prow = quarter * 4 + r;
        const int pcol = n * 16 + sub;
        *reinterpret_cast<__bf16*>(p_wave + prow * 128 + swz(prow, pcol * 2)) =
            (__bf16)p;
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      l_run[r] = l_run[r] * alpha[r] + qsum16(psum[r]);
#pragma unroll
      for (int n = 0; n < 8; ++n) acc_o[n][r] *= alpha[r];
    }
    __syncthreads();

    // ---- O_band += P_band @ V via hardware transpose reads ----
    const unsigned v4_base = (unsigned)(uintptr_t)v4_cur;
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      const int col = n * 16 + sub;
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16x8 ap;
        {
          const int row = sub;
          const int byte_col = (ks * 32 + quarter * 8) * 2;
          ap = *reinterpret_cast<const bf16x8*>(
              p_wave + row * 128 + swz(row, byte_col));
        }
        const int kb0 = (ks * 32 + quarter * 8) >> 2;  // first 4-row tile
        const unsigned a0 = v4_base + v4_addr(kb0 * 4, col);
        const unsigned a1 = v4_base + v4_addr(kb0 * 4 + 4, col);
        u32x2_t r0, r1;
        asm volatile(
            "ds_read_b64_tr_b16 %0, %2\n\t"
            "ds_read_b64_tr_b16 %1, %3\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=v"(r0), "=v"(r1)
            : "v"(a0), "v"(a1));
        u32x4_t bvbits = {r0.x, r0.y, r1.x, r1.y};
        bf16x8 bv = *reinterpret_cast<bf16x8*>(&bvbits);
        acc_o[n] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(ap, bv, acc_o[n],
                                                           0, 0, 0);
      }
    }
    __syncthreads();
  }

  // ---- epilogue (unchanged) ----
  short* o_blk = out + (long long)b * os_b + (long long)h * os_h +
                 (long long)(qt * FA_QT + qrow_w) * os_s;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const float inv = l_run[r] > 0.f ? 1.f / l_run[r] : 0.f;
    const int orow = quarter * 4 + r;
#pragma unroll
    for (int n = 0; n < 8; ++n) {
      o_blk[(long long)orow * os_s + n * 16 + sub] = f2bf(acc_o[n][r] * inv);
    }
    if (sub == 0) {
      const long long lrow =
          ((long long)b * H + h) * S + qt * FA_QT + qrow_w + orow;
      lse[lrow] = m_run[r] + __logf(fmaxf(l_run[r], 1e-30f));
    }
  }
}

extern "C" void flash_attn_fwd_v2_launch(const void* q, const void* k,
                                         const void* v, void* out, void* lse,
                                         int B, int H, int HKV, int S,
                                         float scale, const long long* strides,
                                         hipStream_t stream) {
  dim3 grid(S / FA_QT, H, B);
  hipLaunchKernelGGL(flash_attn_fwd_v2_kernel, grid, dim3(256), 0, stream,
                     (const short*)q, (const short*)k, (const short*)v,
                     (short*)out, (float*)lse, B, H, HKV, S, scale,
                     strides[0], strides[1], strides[2], strides[3],
                     strides[4], strides[5], strides[6], strides[7],
                     strides[8], strides[9], strides[10], strides[11]);
}
