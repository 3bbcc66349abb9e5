// Flash attention forward V3 for MI355X: 32x32x16 MFMA, swapped operands,
// fully in-register online softmax. Supersedes v1 (16x16x32, P staged via
// LDS) per the round-1 plan (profiles/r01e_fa_pmc_findings.txt v3 sketch).
//
// Structure (guide Appendix B "8-warp 32x32 ladder", re-derived for causal
// GQA and verified fragment layouts — tests/test_mfma_gpu.py):
//   block = 512 threads = 8 waves; wave w owns q rows [32w, 32w+32) of a
//   256-row Q block; KV tiles of 64 rows, double-buffered in LDS.
//
//   QK^T SWAPPED: S^T = K @ Q^T via mfma_f32_32x32x16_bf16 with A = K-tile
//   [32 kv x 16 d], B = Q [16 d x 32 q]. C layout (measured, guide §3):
//   col = lane&31 -> ONE q row per lane; row = (reg&3)+8*(reg>>2)+4*(lane>>5)
//   -> 16 kv positions per lane (the lane pair l, l^32 covers 32). The whole
//   softmax row state (m, l) is then ONE scalar per lane — no cross-lane
//   reduction beyond a single permlane32_swap exchange with the partner.
//
//   PV SWAPPED TOO: O^T = V^T @ P^T with A = V^T [32 d x 16 kv] (b128 reads
//   from the swizzled V^T LDS), B = P^T [16 kv x 32 q] — the B fragment is
//   assembled IN REGISTERS from the QK^T output via bf16 packing plus ONE
//   permlane32_swap per register pair (T12). O accumulator: col = q again,
//   so the online-softmax alpha rescale is a per-lane SCALAR multiply.
//
//   LDS: K[64][128] + V^T[128][64], XOR-swizzled, x2 buffers = 64 KB; no P
//   buffer. T14 async-stage: next tile's global loads issue before compute,
//   LDS writes land after the barrier.
//
// The reference (DLRover) has no attention kernel of its own — this is the
// MI355X-native hot path (BASELINE.json north star).
#include <cstdlib>
#include "kern_common.h"

typedef __attribute__((ext_vector_type(8))) __bf16 bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;
typedef __attribute__((ext_vector_type(16))) float f32x16_t;
typedef __attribute__((ext_vector_type(2))) unsigned int u32x2_t;
typedef __attribute__((ext_vector_type(4))) unsigned int u32x4_t;
typedef __attribute__((ext_vector_type(2))) __bf16 bf16x2;

#define FA3_D 128
#define FA3_QB 256  // q rows per block
#define FA3_QW 32   // q rows per wave
#define FA3_KT 64   // kv rows per tile

// XOR swizzle on a byte offset within a 256 B LDS row (guide G4)
__device__ __forceinline__ int swz3(int row, int byte_col) {
  return byte_col ^ ((row & 7) << 4);
}

// pack two fp32 into one u32 of two bf16 (compiler emits v_cvt_pk_bf16_f32)
__device__ __forceinline__ unsigned pack_bf16(float lo, float hi) {
  bf16x2 v = {(__bf16)lo, (__bf16)hi};
  return *reinterpret_cast<unsigned*>(&v);
}

// A-operand fragment X^T[d = n*32 + (l&31)][row = rb + half*8 + i] read from
// a ROW-major swz3-swizzled [rows][128] bf16 LDS tile via two cooperative
// ds_read_b64_tr_b16 gathers. Semantics probe-pinned (profiles/r02a): the
// 16-lane group's addresses form a gather table — lane l supplies the word
// address for (row rb + ((l&15)>>2), d-quad base+(l&3)) and receives, as
// element j, its OWN d column of row rb + j (slots 4j + ((l>>2)&3)).
__device__ __forceinline__ bf16x8 tr_colT_frag(const char* lds, int rb,
                                               int n, int lane, int half) {
  typedef __attribute__((ext_vector_type(2))) unsigned int u32x2_;
  const int j = (lane & 15) >> 2;
  const int dq = n * 8 + ((lane & 16) >> 2) + (lane & 3);
  const int r0 = rb + half * 8 + j;
  const int r1 = r0 + 4;
  const unsigned base = (unsigned)(uintptr_t)lds;
  const unsigned a0 = base + r0 * 256 + ((dq * 8) ^ ((r0 & 7) << 4));
  const unsigned a1 = base + r1 * 256 + ((dq * 8) ^ ((r1 & 7) << 4));
  u32x2_ w0, w1;
  asm volatile(
      "ds_read_b64_tr_b16 %0, %2\n\t"
      "ds_read_b64_tr_b16 %1, %3\n\t"
      "s_waitcnt lgkmcnt(0)"
      : "=v"(w0), "=v"(w1)
      : "v"(a0), "v"(a1));
  u32x4_t frag = u32x4_t{w0.x, w0.y, w1.x, w1.y};
  return *reinterpret_cast<bf16x8*>(&frag);
}

// TRV=false: V staged TRANSPOSED (vt_lds, per-element scatter writes — the
//   measured bank-conflict source) and PV A-operand read b128.
// TRV=true ("v4"): V staged ROW-major like K (conflict-free b128 writes) and
//   the V^T A-operand gathered with ds_read_b64_tr_b16 (tr_colT_frag).
template <bool TRV>
__global__ __launch_bounds__(512) void flash_attn_fwd_v3_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, short* __restrict__ out,
    float* __restrict__ lse, int B, int H, int HKV, int S, float scale,
    long long qs_b, long long qs_h, long long qs_s,
    long long ks_b, long long ks_h, long long ks_s,
    long long vs_b, long long vs_h, long long vs_s,
    long long os_b, long long os_h, long long os_s) {
  __shared__ char k_lds[2][FA3_KT * FA3_D * 2];   // 2 x 16 KB
  __shared__ char vt_lds[2][FA3_D * FA3_KT * 2];  // 2 x 16 KB

  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  const int lane = tid & 63;
  const int half = lane >> 5;   // which 8-wide k-slice of A/B operands
  const int qcol = lane & 31;   // the ONE q row this lane owns

  const int qt = blockIdx.x;
  const int h = blockIdx.y;
  const int b = blockIdx.z;
  const int g = h / (H / HKV);

  const short* q_blk = q + (long long)b * qs_b + (long long)h * qs_h +
                       (long long)(qt * FA3_QB) * qs_s;
  const short* k_head = k + (long long)b * ks_b + (long long)g * ks_h;
  const short* v_head = v + (long long)b * vs_b + (long long)g * vs_h;

  // ---- persistent Q band: B-operand frags, one b128 load each ----
  // frag[ks]: lane holds Q[qrow][ks*16 + half*8 + i], i in [0,8)
  const int q_row_w = wave * FA3_QW + qcol;  // within block
  bf16x8 bq[8];
#pragma unroll
  for (int ks = 0; ks < 8; ++ks) {
    const short* src =
        q_blk + (long long)q_row_w * qs_s + ks * 16 + half * 8;
    bq[ks] = *reinterpret_cast<const bf16x8*>(src);
  }

  const int q_glob = qt * FA3_QB + q_row_w;  // this lane's global q row
  float m_run = -1e30f, l_run = 0.f;
  f32x16_t acc_o[4];
#pragma unroll
  for (int n = 0; n < 4; ++n)
#pragma unroll
    for (int r = 0; r < 16; ++r) acc_o[n][r] = 0.f;

  // staging registers for the async split (T14): each thread carries
  // 2 x b128 of K and 2 x b128 of V for the NEXT tile
  bf16x8 st_k[2], st_v[2];
  // this thread's staging coordinates: thread t covers 16 consecutive
  // elements (2 x b128) of the 64x128 tile
  const int st_row0 = (tid * 16) / FA3_D;      // kv row (0..63)
  const int st_col0 = (tid * 16) % FA3_D;      // d col  (0,16,..,112)

  auto issue_loads = [&](int kt) {
    const short* ksrc = k_head + (long long)(kt * FA3_KT) * ks_s;
    const short* vsrc = v_head + (long long)(kt * FA3_KT) * vs_s;
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      st_k[c] = *reinterpret_cast<const bf16x8*>(
          ksrc + (long long)st_row0 * ks_s + st_col0 + c * 8);
      st_v[c] = *reinterpret_cast<const bf16x8*>(
          vsrc + (long long)st_row0 * vs_s + st_col0 + c * 8);
    }
  };

  auto write_lds = [&](int buf) {
    char* kb = k_lds[buf];
    char* vb = vt_lds[buf];
#pragma unroll
    for (int c = 0; c < 2; ++c) {
      const int col = st_col0 + c * 8;
      *reinterpret_cast<bf16x8*>(kb + st_row0 * 256 + swz3(st_row0, col * 2)) =
          st_k[c];
      if (TRV) {
        // row-major like K: conflict-free b128 store; the transpose happens
        // on READ via the tr_b16 cooperative gather
        *reinterpret_cast<bf16x8*>(vb + st_row0 * 256 +
                                   swz3(st_row0, col * 2)) = st_v[c];
      } else {
        // V transposed: element (row, col+i) -> vt[col+i][row]
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          const int trow = col + i;
          *reinterpret_cast<__bf16*>(vb + trow * 128 +
                                     swz3(trow, st_row0 * 2)) = st_v[c][i];
        }
      }
    }
  };

  const int n_tiles = (qt + 1) * (FA3_QB / FA3_KT);  // causal upper bound
  // this wave's last needed tile (q rows < its band contribute nothing)
  const int wave_q_max = qt * FA3_QB + wave * FA3_QW + FA3_QW - 1;

  issue_loads(0);
  write_lds(0);
  __syncthreads();

  for (int kt = 0; kt < n_tiles; ++kt) {
    const int cur = kt & 1;
    const char* k_cur = k_lds[cur];
    const char* vt_cur = vt_lds[cur];
    if (kt + 1 < n_tiles) issue_loads(kt + 1);  // T14: issue early

    const int kv0 = kt * FA3_KT;
    const bool wave_active = kv0 <= wave_q_max;

    if (wave_active) {
      // ---- S^T tile: two 32x32 outputs (kv subtiles), K=128 ----
      f32x16_t acc_s[2];
#pragma unroll
      for (int n = 0; n < 2; ++n)
#pragma unroll
        for (int r = 0; r < 16; ++r) acc_s[n][r] = 0.f;
#pragma unroll
      for (int n = 0; n < 2; ++n) {
#pragma unroll
        for (int ks = 0; ks < 8; ++ks) {
          // A = K[kv = n*32 + (lane&31)][d = ks*16 + half*8 + i]
          bf16x8 ak;
          {
            const int row = n * 32 + qcol;
            const int byte_col = (ks * 16 + half * 8) * 2;
            ak = *reinterpret_cast<const bf16x8*>(
                k_cur + row * 256 + swz3(row, byte_col));
          }
          acc_s[n] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ak, bq[ks],
                                                             acc_s[n], 0, 0, 0);
        }
      }

      // ---- causal mask + scale; per-lane online softmax ----
      // lane holds kv rows (r&3)+8*(r>>2)+4*half (+32 for acc_s[1])
      float pmax = -1e30f;
#pragma unroll
      for (int n = 0; n < 2; ++n) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int kv_glob =
              kv0 + n * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          float s = acc_s[n][r] * scale;
          if (kv_glob > q_glob) s = -1e30f;
          acc_s[n][r] = s;
          pmax = fmaxf(pmax, s);
        }
      }
      {
        // combine with the partner lane (same q, other 32 kv rows)
        u32x2_t sw = __builtin_amdgcn_permlane32_swap(
            __builtin_bit_cast(unsigned, pmax),
            __builtin_bit_cast(unsigned, pmax), false, false);
        const float other = __builtin_bit_cast(
            float, half ? sw.x : sw.y);
        pmax = fmaxf(pmax, other);
      }
      const float m_new = fmaxf(m_run, pmax);
      const float alpha = __expf(m_run - m_new);
      m_run = m_new;

      float psum = 0.f;
      float p[32];
#pragma unroll
      for (int n = 0; n < 2; ++n) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const float e = __expf(acc_s[n][r] - m_new);
          p[n * 16 + r] = e;
          psum += e;
        }
      }
      {
        u32x2_t sw = __builtin_amdgcn_permlane32_swap(
            __builtin_bit_cast(unsigned, psum),
            __builtin_bit_cast(unsigned, psum), false, false);
        psum += __builtin_bit_cast(float, half ? sw.x : sw.y);
      }
      l_run = l_run * alpha + psum;
#pragma unroll
      for (int n = 0; n < 4; ++n)
#pragma unroll
        for (int r = 0; r < 16; ++r) acc_o[n][r] *= alpha;

      // ---- assemble P^T B-frags: pack pairs to bf16, ONE swap per pair ---
      // own regs r, r+1 = kv (base + 0,1); partner holds (base + 4,5).
      // After permlane32_swap(a=own pair, b=own pair+4):
      //   half 0 lanes: a' = kv(+0,1), b' = partner kv(+4,5)
      //   half 1 lanes: a' = partner's,  b (own) = kv(+12,13) ...
      // yielding for every lane the 8 kv slots (half*8 .. half*8+8) of each
      // 16-kv k-slice, i.e. exactly the B fragment.
      bf16x8 bp[4];  // k-slices: [subtile n][ks within 32] = n*2 + ks
#pragma unroll
      for (int n = 0; n < 2; ++n) {
#pragma unroll
        for (int ks2 = 0; ks2 < 2; ++ks2) {
          // k-slice covers kv = n*32 + ks2*16 + [0,16)
          // own regs for this slice: r0 = ks2*8 (kv base n*32+ks2*16+4*half)
          const int r0 = n * 16 + ks2 * 8;
          unsigned w0 = pack_bf16(p[r0 + 0], p[r0 + 1]);
          unsigned w1 = pack_bf16(p[r0 + 2], p[r0 + 3]);
          unsigned w2 = pack_bf16(p[r0 + 4], p[r0 + 5]);
          unsigned w3 = pack_bf16(p[r0 + 6], p[r0 + 7]);
          // swap(vdst, vsrc) exchanges vdst's lanes 32-63 with vsrc's 0-31:
          //   .x (new w0): h0 lanes own kv(+0,1); h1 lanes partner kv(+8,9)
          //   .y (new w2): h0 lanes partner kv(+4,5); h1 lanes own kv(+12,13)
          // -> frag order [x0, x1, y0, y1] is the SAME for both halves
          u32x2_t s0 = __builtin_amdgcn_permlane32_swap(w0, w2, false, false);
          u32x2_t s1 = __builtin_amdgcn_permlane32_swap(w1, w3, false, false);
          u32x4_t frag = u32x4_t{s0.x, s1.x, s0.y, s1.y};
          bp[n * 2 + ks2] = *reinterpret_cast<bf16x8*>(&frag);
        }
      }

      // ---- O^T += V^T @ P^T ----
#pragma unroll
      for (int n = 0; n < 4; ++n) {  // d blocks of 32
#pragma unroll
        for (int ks = 0; ks < 4; ++ks) {  // kv k-slices of 16
          // A = V^T[d = n*32 + (lane&31)][kv = ks*16 + half*8 + i]
          bf16x8 av;
          if (TRV) {
            av = tr_colT_frag(vt_cur, ks * 16, n, lane, half);
          } else {
            const int row = n * 32 + qcol;
            const int byte_col = (ks * 16 + half * 8) * 2;
            av = *reinterpret_cast<const bf16x8*>(
                vt_cur + row * 128 + swz3(row, byte_col));
          }
          acc_o[n] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(av, bp[ks],
                                                             acc_o[n], 0, 0, 0);
        }
      }
    }

    __syncthreads();  // all reads of buf[cur] done
    if (kt + 1 < n_tiles) {
      write_lds(cur ^ 1);  // T14: write late (loads have landed by now)
      __syncthreads();
    }
  }

  // ---- epilogue: normalize, bounce through LDS, coalesced b128 stores ----
  // O^T register layout: lane owns q = q_row_w, d = n*32+(r&3)+8*(r>>2)+4h —
  // direct global stores would be 2-byte scatters. The kv-loop LDS is free
  // now: 4 waves at a time write their 32x128 band into an 8 KB swizzled
  // slab, then read it back row-major and store b128.
  const float inv = l_run > 0.f ? 1.f / l_run : 0.f;
  short* o_blk = out + (long long)b * os_b + (long long)h * os_h +
                 (long long)(qt * FA3_QB) * os_s;
  char* slab = (char*)k_lds + (wave & 3) * (FA3_QW * FA3_D * 2);
  for (int round = 0; round < 2; ++round) {
    __syncthreads();
    if ((wave >> 2) == round) {
#pragma unroll
      for (int n = 0; n < 4; ++n) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int d = n * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          *reinterpret_cast<__bf16*>(slab + qcol * 256 + swz3(qcol, d * 2)) =
              (__bf16)(acc_o[n][r] * inv);
        }
      }
    }
    __syncthreads();
    if ((wave >> 2) == round) {
      // 64 lanes x 16 elems = 2 rows of 128 per pass; 32-row band -> 4
      // passes of 2 x b128 per lane
#pragma unroll
      for (int pass = 0; pass < 4; ++pass) {
        const int elem0 = (pass * 64 + lane) * 16;
        const int r2 = elem0 / FA3_D;
        const int c2 = elem0 % FA3_D;
        bf16x8 v0 = *reinterpret_cast<const bf16x8*>(
            slab + r2 * 256 + swz3(r2, c2 * 2));
        bf16x8 v1 = *reinterpret_cast<const bf16x8*>(
            slab + r2 * 256 + swz3(r2, (c2 + 8) * 2));
        short* dst = o_blk + (long long)(wave * FA3_QW + r2) * os_s + c2;
        *reinterpret_cast<bf16x8*>(dst) = v0;
        *reinterpret_cast<bf16x8*>(dst + 8) = v1;
      }
    }
  }
  if (half == 0 && lse != nullptr) {
    const long long lrow = ((long long)b * H + h) * S + q_glob;
    lse[lrow] = m_run + __logf(fmaxf(l_run, 1e-30f));
  }
}

extern "C" void flash_attn_fwd_v3_launch(const void* q, const void* k,
                                         const void* v, void* out, void* lse,
                                         int B, int H, int HKV, int S,
                                         float scale, const long long* strides,
                                         hipStream_t stream) {
  dim3 grid(S / FA3_QB, H, B);
  // v4 (tr-read V, conflict-free staging) unless DLROVER_FA_TRV=0
  static const bool trv = []() {
    const char* e = getenv("DLROVER_FA_TRV");
    return e == nullptr || e[0] != '0';
  }();
  if (trv) {
    hipLaunchKernelGGL((flash_attn_fwd_v3_kernel<true>), grid, dim3(512), 0,
                       stream, (const short*)q, (const short*)k,
                       (const short*)v, (short*)out, (float*)lse, B, H, HKV,
                       S, scale, strides[0], strides[1], strides[2],
                       strides[3], strides[4], strides[5], strides[6],
                       strides[7], strides[8], strides[9], strides[10],
                       strides[11]);
  } else {
    hipLaunchKernelGGL((flash_attn_fwd_v3_kernel<false>), grid, dim3(512), 0,
                       stream, (const short*)q, (const short*)k,
                       (const short*)v, (short*)out, (float*)lse, B, H, HKV,
                       S, scale, strides[0], strides[1], strides[2],
                       strides[3], strides[4], strides[5], strides[6],
                       strides[7], strides[8], strides[9], strides[10],
                       strides[11]);
  }
}
