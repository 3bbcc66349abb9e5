// Python bindings for the dlrover_amd HIP kernels.
//
// Host-only translation layer: validates tensors, allocates outputs through
// the torch caching allocator, and forwards raw pointers + the current HIP
// stream to the extern "C" launchers in the .hip translation units.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>

#include <tuple>

extern "C" {
void rmsnorm_fwd_launch(const void*, const void*, void*, void*, int, int,
                        float, void*);
void rmsnorm_add_fwd_launch(const void*, const void*, const void*, void*,
                            void*, void*, int, int, float, void*);
void rmsnorm_bwd_launch(const void*, const void*, const void*, const void*,
                        const void*, void*, void*, void*, void*, int, int,
                        int, void*);
void rope_launch(void*, const void*, const void*, const void*, long long,
                 long long, int, int, int, void*);
void rope_oop_launch(const void*, void*, const void*, const void*,
                     const void*, long long, long long, int, int, long long,
                     int, void*);
void swiglu_fwd_launch(const void*, void*, long long, int, void*);
void swiglu_bwd_launch(const void*, const void*, void*, long long, int, void*);
void adamw_launch(void*, const void*, void*, void*, void*, long long, float,
                  float, float, float, float, int, int, float, void*);
void causal_softmax_fwd_launch(void*, long long, int, int, int, float, void*);
void causal_softmax_bwd_launch(void*, const void*, long long, int, float,
                               void*);
void cross_entropy_launch(void*, const void*, void*, long long, int, int,
                          float, int, void*);
void mfma16_probe_launch(const void*, const void*, void*, void*);
void mfma32_probe_launch(const void*, const void*, void*, void*);
void tr_b16_probe3_launch(void*, void*);
void tr_b16_probe_launch(void*, void*);
void flash_attn_fwd_launch(const void*, const void*, const void*, void*,
                           void*, int, int, int, int, float,
                           const long long*, void*);
void flash_attn_fwd_v3_launch(const void*, const void*, const void*, void*,
                              void*, int, int, int, int, float,
                              const long long*, void*);
void flash_attn_fwd_v2_launch(const void*, const void*, const void*, void*,
                              void*, int, int, int, int, float,
                              const long long*, void*);
void fa_bwd_pre_launch(const void*, const void*, void*, int, int, int,
                       const long long*, const long long*, void*);
void fa_bwd_dq_launch(const void*, const void*, const void*, const void*,
                      const void*, const void*, void*, int, int, int, int,
                      float, const long long*, void*);
void fa_bwd_dq_v3_launch(const void*, const void*, const void*, const void*,
                         const void*, const void*, void*, int, int, int, int,
                         float, const long long*, void*);
void fa_bwd_dkv_v3_launch(const void*, const void*, const void*, const void*,
                          const void*, const void*, void*, void*, int, int,
                          int, int, float, const long long*, void*);
void fa_bwd_dkv_launch(const void*, const void*, const void*, const void*,
                       const void*, const void*, void*, void*, int, int, int,
                       int, float, const long long*, void*);
void f32_to_bf16_launch(const void*, void*, long long, void*);
}

namespace {

void* cur_stream() {
  return (void*)at::cuda::getCurrentCUDAStream().stream();
}

void check_bf16(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == at::kBFloat16, name, " must be bf16");
}

std::tuple<at::Tensor, at::Tensor> rmsnorm_fwd(const at::Tensor& x,
                                               const at::Tensor& w,
                                               double eps) {
  check_bf16(x, "x");
  check_bf16(w, "w");
  const int hidden = (int)x.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  const long long n_rows = x.numel() / hidden;
  auto y = at::empty_like(x);
  auto invrms = at::empty({n_rows}, x.options().dtype(at::kFloat));
  rmsnorm_fwd_launch(x.data_ptr(), w.data_ptr(), y.data_ptr(),
                     invrms.data_ptr(), (int)n_rows, hidden, (float)eps,
                     cur_stream());
  return {y, invrms};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> rmsnorm_add_fwd(
    const at::Tensor& x, const at::Tensor& resid, const at::Tensor& w,
    double eps) {
  check_bf16(x, "x");
  check_bf16(resid, "resid");
  const int hidden = (int)x.size(-1);
  const long long n_rows = x.numel() / hidden;
  auto h = at::empty_like(x);
  auto y = at::empty_like(x);
  auto invrms = at::empty({n_rows}, x.options().dtype(at::kFloat));
  rmsnorm_add_fwd_launch(x.data_ptr(), resid.data_ptr(), w.data_ptr(),
                         h.data_ptr(), y.data_ptr(), invrms.data_ptr(),
                         (int)n_rows, hidden, (float)eps, cur_stream());
  return {h, y, invrms};
}

std::tuple<at::Tensor, at::Tensor> rmsnorm_bwd(
    const at::Tensor& dy, const at::Tensor& x, const at::Tensor& w,
    const at::Tensor& invrms, c10::optional<at::Tensor> dh_extra) {
  check_bf16(dy, "dy");
  check_bf16(x, "x");
  const int hidden = (int)x.size(-1);
  TORCH_CHECK(hidden <= 16384, "rmsnorm_bwd: hidden > 16384 unsupported "
              "(register dw accumulator: RMSN_MAX_VPT)");
  const long long n_rows = x.numel() / hidden;
  // one private slice per block (plain stores, no atomics): n_partials must
  // match the kernel grid = min(n_rows, 2048)
  const int n_partials = (int)(n_rows < 2048 ? (n_rows > 0 ? n_rows : 1) : 2048);
  auto dx = at::empty_like(x);
  auto dw = at::empty_like(w);
  auto partial =
      at::empty({n_partials, hidden}, x.options().dtype(at::kFloat));
  auto dw_f32 = at::zeros({hidden}, x.options().dtype(at::kFloat));
  const void* dh_ptr = nullptr;
  if (dh_extra.has_value()) {
    check_bf16(*dh_extra, "dh_extra");
    dh_ptr = dh_extra->data_ptr();
  }
  rmsnorm_bwd_launch(dy.data_ptr(), x.data_ptr(), w.data_ptr(),
                     invrms.data_ptr(), dh_ptr, dx.data_ptr(),
                     partial.data_ptr(), dw_f32.data_ptr(), dw.data_ptr(),
                     (int)n_rows, hidden, n_partials, cur_stream());
  return {dx, dw};
}

void rope_apply(at::Tensor& x, const at::Tensor& pos, const at::Tensor& cos,
                const at::Tensor& sin, bool backward) {
  check_bf16(x, "x");
  TORCH_CHECK(pos.scalar_type() == at::kInt && pos.is_cuda(),
              "pos must be int32 on GPU");
  TORCH_CHECK(cos.scalar_type() == at::kFloat && cos.is_contiguous(),
              "cos table must be contiguous fp32");
  const int head_dim = (int)x.size(-1);
  const int n_heads = (int)x.size(-2);
  TORCH_CHECK(head_dim % 8 == 0, "head_dim must be a multiple of 8");
  const long long n_tokens = x.numel() / ((long long)n_heads * head_dim);
  TORCH_CHECK(pos.numel() > 0 && n_tokens % pos.numel() == 0,
              "pos length must divide the token count (one entry per "
              "sequence position)");
  rope_launch(x.data_ptr(), pos.data_ptr(), cos.data_ptr(), sin.data_ptr(),
              n_tokens, (long long)pos.numel(), n_heads, head_dim,
              backward ? 1 : 0, cur_stream());
}

at::Tensor rope_rotate_oop(const at::Tensor& x, const at::Tensor& pos,
                           const at::Tensor& cos, const at::Tensor& sin,
                           bool backward) {
  // x: [..., n_tokens?, n_heads, head_dim] with heads*head_dim contiguous
  // per token row; the token stride may exceed n_heads*head_dim (a view out
  // of the fused QKV projection). Returns a fresh CONTIGUOUS tensor.
  TORCH_CHECK(x.is_cuda(), "x must be on GPU");
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "x must be bf16");
  TORCH_CHECK(pos.scalar_type() == at::kInt && pos.is_cuda(),
              "pos must be int32 on GPU");
  const int head_dim = (int)x.size(-1);
  const int n_heads = (int)x.size(-2);
  TORCH_CHECK(head_dim % 8 == 0, "head_dim must be a multiple of 8");
  TORCH_CHECK(x.stride(-1) == 1 && x.stride(-2) == head_dim,
              "rope: head_dim/heads must be contiguous within a token row");
  const long long n_tokens = x.numel() / ((long long)n_heads * head_dim);
  // flatten every leading dim into tokens: requires uniform token stride
  long long tok_stride = n_heads * (long long)head_dim;
  if (x.dim() >= 3) {
    tok_stride = x.stride(-3);
    long long rows = x.size(-3);
    for (int d = (int)x.dim() - 4; d >= 0; --d) {
      TORCH_CHECK(x.stride(d) == rows * tok_stride,
                  "rope: leading dims must be uniformly strided");
      rows *= x.size(d);
    }
  }
  TORCH_CHECK(pos.numel() > 0 && n_tokens % pos.numel() == 0,
              "pos length must divide the token count");
  auto out = at::empty(x.sizes(), x.options());
  rope_oop_launch(x.data_ptr(), out.data_ptr(), pos.data_ptr(),
                  cos.data_ptr(), sin.data_ptr(), n_tokens,
                  (long long)pos.numel(), n_heads, head_dim, tok_stride,
                  backward ? 1 : 0, cur_stream());
  return out;
}

at::Tensor swiglu_fwd(const at::Tensor& gate_up) {
  check_bf16(gate_up, "gate_up");
  const int two_inner = (int)gate_up.size(-1);
  TORCH_CHECK(two_inner % 16 == 0, "inner dim must be a multiple of 8");
  const int inner = two_inner / 2;
  const long long n_rows = gate_up.numel() / two_inner;
  auto sizes = gate_up.sizes().vec();
  sizes.back() = inner;
  auto out = at::empty(sizes, gate_up.options());
  swiglu_fwd_launch(gate_up.data_ptr(), out.data_ptr(), n_rows, inner,
                    cur_stream());
  return out;
}

at::Tensor swiglu_bwd(const at::Tensor& dy, const at::Tensor& gate_up) {
  check_bf16(dy, "dy");
  check_bf16(gate_up, "gate_up");
  const int inner = (int)gate_up.size(-1) / 2;
  const long long n_rows = gate_up.numel() / (2LL * inner);
  auto d_gate_up = at::empty_like(gate_up);
  swiglu_bwd_launch(dy.data_ptr(), gate_up.data_ptr(), d_gate_up.data_ptr(),
                    n_rows, inner, cur_stream());
  return d_gate_up;
}

void adamw_step(at::Tensor& param, const at::Tensor& grad, at::Tensor& m,
                at::Tensor& v, c10::optional<at::Tensor> param_bf16,
                double lr, double beta1, double beta2, double eps,
                double weight_decay, int64_t step, double grad_scale) {
  TORCH_CHECK(param.scalar_type() == at::kFloat && param.is_contiguous(),
              "master param must be contiguous fp32");
  TORCH_CHECK(grad.is_cuda() && grad.is_contiguous(), "grad must be GPU");
  TORCH_CHECK(grad.numel() == param.numel(), "grad/param numel mismatch");
  const bool gbf16 = grad.scalar_type() == at::kBFloat16;
  TORCH_CHECK(gbf16 || grad.scalar_type() == at::kFloat,
              "grad must be bf16 or fp32");
  void* pb = nullptr;
  if (param_bf16.has_value()) {
    check_bf16(*param_bf16, "param_bf16");
    pb = param_bf16->data_ptr();
  }
  adamw_launch(param.data_ptr(), grad.data_ptr(), m.data_ptr(), v.data_ptr(),
               pb, param.numel(), (float)lr, (float)beta1, (float)beta2,
               (float)eps, (float)weight_decay, (int)step, gbf16 ? 1 : 0,
               (float)grad_scale, cur_stream());
}

void causal_softmax_fwd(at::Tensor& scores, int64_t q_len, int64_t q_offset,
                        double scale) {
  check_bf16(scores, "scores");
  const int row_len = (int)scores.size(-1);
  TORCH_CHECK(row_len % 8 == 0, "row_len must be a multiple of 8");
  const long long n_rows = scores.numel() / row_len;
  causal_softmax_fwd_launch(scores.data_ptr(), n_rows, row_len, (int)q_len,
                            (int)q_offset, (float)scale, cur_stream());
}

void causal_softmax_bwd(at::Tensor& dscores, const at::Tensor& probs,
                        double scale) {
  check_bf16(dscores, "dscores");
  check_bf16(probs, "probs");
  const int row_len = (int)dscores.size(-1);
  const long long n_rows = dscores.numel() / row_len;
  causal_softmax_bwd_launch(dscores.data_ptr(), probs.data_ptr(), n_rows,
                            row_len, (float)scale, cur_stream());
}

// q/k/v/out are LOGICALLY [B,H,S,D]; strided permuted views are fine as long
// as the last (D) dim is contiguous — no transpose copies needed.
static void check_attn_view(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kBFloat16, name,
              " must be bf16 GPU");
  TORCH_CHECK(t.dim() == 4 && t.stride(3) == 1, name,
              " must be [B,H,S,D] with contiguous D");
}

static void pack_strides(long long* st, const at::Tensor& t, int at) {
  st[at + 0] = (long long)t.stride(0);
  st[at + 1] = (long long)t.stride(1);
  st[at + 2] = (long long)t.stride(2);
}

std::tuple<at::Tensor, at::Tensor> flash_attn_fwd(const at::Tensor& q,
                                                  const at::Tensor& k,
                                                  const at::Tensor& v,
                                                  double scale) {
  check_attn_view(q, "q");
  check_attn_view(k, "k");
  check_attn_view(v, "v");
  const int B = (int)q.size(0), H = (int)q.size(1), S = (int)q.size(2);
  const int D = (int)q.size(3), HKV = (int)k.size(1);
  TORCH_CHECK(D == 128, "flash_attn: head_dim must be 128");
  TORCH_CHECK(S % 64 == 0, "flash_attn: seq len must be a multiple of 64");
  TORCH_CHECK(H % HKV == 0, "flash_attn: H % HKV != 0");
  // output in [B,S,H,D] storage (what the model consumes — avoids the
  // transpose-back copy); returned as a [B,H,S,D] view
  auto out_bshd = at::empty({B, S, H, D}, q.options());
  auto out = out_bshd.permute({0, 2, 1, 3});
  auto lse = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  long long st[12];
  pack_strides(st, q, 0);
  pack_strides(st, k, 3);
  pack_strides(st, v, 6);
  pack_strides(st, out, 9);
  // DLROVER_FA_V2=1 selects the experimental ds_read_b64_tr_b16 PV path.
  // Hardware measurement showed the tr read takes its tile-selecting
  // address bits from the 16-lane GROUP LEADER (only bits 1-2 act
  // per-lane), so the per-lane-tile v2 design reads wrong columns for
  // sub>3 — kept off until the round-2 redesign (see attention.hip).
  static const bool use_v2 = []() {
    const char* e = getenv("DLROVER_FA_V2");
    return e != nullptr && e[0] == '1';
  }();
  // V3 (default when shapes allow): 32x32x16 MFMA, swapped operands,
  // in-register online softmax — attention_v3.hip. DLROVER_FA_V3=0 falls
  // back to v1 (16x16x32 + LDS-staged P).
  static const bool v3_enabled = []() {
    const char* e = getenv("DLROVER_FA_V3");
    return e == nullptr || e[0] != '0';
  }();
  if (v3_enabled && !use_v2 && S % 256 == 0) {
    flash_attn_fwd_v3_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                             out_bshd.data_ptr(), lse.data_ptr(), B, H, HKV,
                             S, (float)scale, st, cur_stream());
  } else if (use_v2) {
    flash_attn_fwd_v2_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                             out_bshd.data_ptr(), lse.data_ptr(), B, H, HKV,
                             S, (float)scale, st, cur_stream());
  } else {
    flash_attn_fwd_launch(q.data_ptr(), k.data_ptr(), v.data_ptr(),
                          out_bshd.data_ptr(), lse.data_ptr(), B, H, HKV, S,
                          (float)scale, st, cur_stream());
  }
  return {out, lse};
}

std::tuple<at::Tensor, at::Tensor, at::Tensor> flash_attn_bwd(
    const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
    const at::Tensor& out, const at::Tensor& dout_in, const at::Tensor& lse,
    double scale) {
  check_attn_view(q, "q");
  check_attn_view(k, "k");
  check_attn_view(v, "v");
  const int B = (int)q.size(0), H = (int)q.size(1), S = (int)q.size(2);
  const int D = (int)q.size(3), HKV = (int)k.size(1);
  // the preprocess + dkv stage read dO/O with [B,S,H,D]-style strides; accept
  // any incoming grad layout by normalizing dO into out's layout
  at::Tensor dout = dout_in;
  if (dout.strides() != out.strides()) {
    dout = at::empty_like(out.permute({0, 2, 1, 3}).contiguous())
               .view({B, S, H, D});
    dout = dout.permute({0, 2, 1, 3});
    dout.copy_(dout_in);
  }
  TORCH_CHECK(dout.stride(3) == 1, "dout needs contiguous D");
  auto dvec = at::empty({B, H, S}, q.options().dtype(at::kFloat));
  {
    long long ost[3], dstr[3];
    pack_strides(ost, out, 0);
    pack_strides(dstr, dout, 0);
    fa_bwd_pre_launch(dout.data_ptr(), out.data_ptr(), dvec.data_ptr(), B, H,
                      S, ost, dstr, cur_stream());
    long long st[12];
    pack_strides(st, q, 0);
    pack_strides(st, k, 3);
    pack_strides(st, v, 6);
    pack_strides(st, dout, 9);
    // dq mirrors q's layout: allocate [B,S,H,D] storage, return as view
    auto dq_bshd = at::empty({B, S, H, D}, q.options());
    auto dq = dq_bshd.permute({0, 2, 1, 3});
    {
      long long stq[12];
      pack_strides(stq, dq, 0);
      pack_strides(stq, k, 3);
      pack_strides(stq, v, 6);
      pack_strides(stq, dout, 9);
      // q-strides slot is used for BOTH q loads and dq stores in the kernel;
      // they must match, so stage q into dq's layout if they differ
      at::Tensor quse = q;
      if (q.strides() != dq.strides()) {
        auto q_bshd = at::empty({B, S, H, D}, q.options());
        auto qv = q_bshd.permute({0, 2, 1, 3});
        qv.copy_(q);
        quse = qv;
        pack_strides(stq, quse, 0);
      }
      // v3 backward (32x32 MFMA swapped rebuild) when shapes allow;
      // DLROVER_FA_BWD_V3=0 reverts to the 16x16 kernels
      static const bool bwd_v3 = []() {
        const char* e = getenv("DLROVER_FA_BWD_V3");
        return e == nullptr || e[0] != '0';
      }();
      auto dk32 = at::zeros({B, HKV, S, D}, k.options().dtype(at::kFloat));
      auto dv32 = at::zeros({B, HKV, S, D}, v.options().dtype(at::kFloat));
      if (bwd_v3 && S % 256 == 0) {
        fa_bwd_dq_v3_launch(quse.data_ptr(), k.data_ptr(), v.data_ptr(),
                            dout.data_ptr(), lse.data_ptr(), dvec.data_ptr(),
                            dq_bshd.data_ptr(), B, H, HKV, S, (float)scale,
                            stq, cur_stream());
        fa_bwd_dkv_v3_launch(quse.data_ptr(), k.data_ptr(), v.data_ptr(),
                             dout.data_ptr(), lse.data_ptr(),
                             dvec.data_ptr(), dk32.data_ptr(),
                             dv32.data_ptr(), B, H, HKV, S, (float)scale,
                             stq, cur_stream());
      } else {
        fa_bwd_dq_launch(quse.data_ptr(), k.data_ptr(), v.data_ptr(),
                         dout.data_ptr(), lse.data_ptr(), dvec.data_ptr(),
                         dq_bshd.data_ptr(), B, H, HKV, S, (float)scale, stq,
                         cur_stream());
        fa_bwd_dkv_launch(quse.data_ptr(), k.data_ptr(), v.data_ptr(),
                          dout.data_ptr(), lse.data_ptr(), dvec.data_ptr(),
                          dk32.data_ptr(), dv32.data_ptr(), B, H, HKV, S,
                          (float)scale, stq, cur_stream());
      }
      auto dk = at::empty({B, HKV, S, D}, k.options());
      auto dv = at::empty({B, HKV, S, D}, v.options());
      f32_to_bf16_launch(dk32.data_ptr(), dk.data_ptr(), dk32.numel(),
                         cur_stream());
      f32_to_bf16_launch(dv32.data_ptr(), dv.data_ptr(), dv32.numel(),
                         cur_stream());
      return {dq, dk, dv};
    }
  }
}

at::Tensor mfma16_probe(const at::Tensor& A, const at::Tensor& B) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  TORCH_CHECK(A.sizes() == at::IntArrayRef({16, 32}) &&
              B.sizes() == at::IntArrayRef({32, 16}),
              "probe wants A[16,32], B[32,16]");
  auto C = at::empty({16, 16}, A.options().dtype(at::kFloat));
  mfma16_probe_launch(A.data_ptr(), B.data_ptr(), C.data_ptr(), cur_stream());
  return C;
}

// Direct async copies between device tensors and FOREIGN host memory
// (hipHostRegister'ed shm mappings). torch's copy_ cannot be trusted here:
// it may not recognize the registered pointer as pinned and fall back to a
// synchronous null-stream hipMemcpy, which serializes with ALL compute on
// the default stream (measured: each 104-GB ckpt drain added its full ~2 s
// to the training wall instead of overlapping). These run on the CURRENT
// torch stream — call inside `with torch.cuda.stream(side)`.
void memcpy_d2h_async(int64_t dst_addr, const at::Tensor& src) {
  TORCH_CHECK(src.is_cuda() && src.is_contiguous(), "src must be cuda+contig");
  auto err = hipMemcpyAsync((void*)dst_addr, src.data_ptr(), src.nbytes(),
                            hipMemcpyDeviceToHost,
                            (hipStream_t)cur_stream());
  TORCH_CHECK(err == hipSuccess, "hipMemcpyAsync D2H: ", hipGetErrorString(err));
}

void memcpy_h2d_async(at::Tensor& dst, int64_t src_addr, int64_t nbytes) {
  TORCH_CHECK(dst.is_cuda() && dst.is_contiguous(), "dst must be cuda+contig");
  TORCH_CHECK(nbytes <= (int64_t)dst.nbytes(), "overflow");
  auto err = hipMemcpyAsync(dst.data_ptr(), (const void*)src_addr, nbytes,
                            hipMemcpyHostToDevice,
                            (hipStream_t)cur_stream());
  TORCH_CHECK(err == hipSuccess, "hipMemcpyAsync H2D: ", hipGetErrorString(err));
}

at::Tensor mfma32_probe(const at::Tensor& A, const at::Tensor& B) {
  check_bf16(A, "A");
  check_bf16(B, "B");
  TORCH_CHECK(A.sizes() == at::IntArrayRef({32, 16}) &&
              B.sizes() == at::IntArrayRef({16, 32}),
              "probe wants A[32,16], B[16,32]");
  auto C = at::empty({32, 32}, A.options().dtype(at::kFloat));
  mfma32_probe_launch(A.data_ptr(), B.data_ptr(), C.data_ptr(), cur_stream());
  return C;
}

at::Tensor cross_entropy_fwd_bwd(at::Tensor& logits, const at::Tensor& targets,
                                 int64_t ignore_index, double grad_scale,
                                 bool compute_grad) {
  check_bf16(logits, "logits");
  TORCH_CHECK(targets.scalar_type() == at::kInt && targets.is_cuda(),
              "targets must be int32 on GPU");
  const int vocab = (int)logits.size(-1);
  const long long n_rows = logits.numel() / vocab;
  TORCH_CHECK(targets.numel() == n_rows, "one target per row");
  auto losses = at::empty({n_rows}, logits.options().dtype(at::kFloat));
  cross_entropy_launch(logits.data_ptr(), targets.data_ptr(),
                       losses.data_ptr(), n_rows, vocab, (int)ignore_index,
                       (float)grad_scale, compute_grad ? 1 : 0, cur_stream());
  return losses;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "dlrover_amd CDNA4 (gfx950) kernels";
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "fused RMSNorm forward (bf16)");
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "fused RMSNorm backward (bf16)");
  m.def("rmsnorm_add_fwd", &rmsnorm_add_fwd,
        "fused residual-add + RMSNorm forward -> (h, y, invrms)");
  m.def("rope_apply", &rope_apply, "in-place RoPE rotate-half (bf16)");
  m.def("rope_rotate_oop", &rope_rotate_oop,
        "out-of-place RoPE: strided source -> contiguous output");
  m.def("swiglu_fwd", &swiglu_fwd, "fused SwiGLU forward (bf16)");
  m.def("swiglu_bwd", &swiglu_bwd, "fused SwiGLU backward (bf16)");
  m.def("adamw_step", &adamw_step, "fused AdamW with fp32 master weights");
  m.def("causal_softmax_fwd", &causal_softmax_fwd,
        "in-place fused scale+causal-mask+softmax (bf16)");
  m.def("causal_softmax_bwd", &causal_softmax_bwd,
        "in-place softmax backward (bf16)");
  m.def("cross_entropy_fwd_bwd", &cross_entropy_fwd_bwd,
        "fused CE loss + in-place dlogits (bf16)");
  m.def("tr_b16_probe", []() {
    auto out = at::empty({64, 12}, at::TensorOptions()
                                      .dtype(at::kBFloat16)
                                      .device(at::kCUDA));
    tr_b16_probe_launch(out.data_ptr(), cur_stream());
    return out;
  }, "ds_read_b64_tr_b16 semantics probe");
  m.def("mfma16_probe", &mfma16_probe, "MFMA 16x16x32 bf16 layout self-test");
  m.def("mfma32_probe", &mfma32_probe, "MFMA 32x32x16 bf16 layout self-test");
  m.def("memcpy_d2h_async", &memcpy_d2h_async,
        "hipMemcpyAsync device tensor -> registered host address");
  m.def("memcpy_h2d_async", &memcpy_h2d_async,
        "hipMemcpyAsync registered host address -> device tensor");
  m.def("tr_b16_probe3", []() {
    auto out = at::empty({64, 12}, at::TensorOptions()
                                      .dtype(at::kShort)
                                      .device(at::kCUDA));
    tr_b16_probe3_launch(out.data_ptr(), cur_stream());
    return out;
  }, "tr_b16 cooperative tile-sourcing probe (raw index bits)");
  m.def("flash_attn_fwd", &flash_attn_fwd,
        "flash attention forward (bf16, causal, GQA, D=128) -> (out, lse)");
  m.def("flash_attn_bwd", &flash_attn_bwd,
        "flash attention backward -> (dq, dk, dv)");
}
