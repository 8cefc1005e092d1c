// Python bindings for the MI355X flow-GNN kernels (deepdfa_amd._C).

#include <torch/extension.h>

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <c10/hip/HIPStream.h>

// the launchers are defined (at global scope) in flowgnn_kernels.hip
template <typename T>
void launch_embed4_fwd(const T*, const long*, T*, int, int, hipStream_t);
template <typename T, typename TO>
void launch_embed4_fwd2(const T*, const long*, TO*, int, int, hipStream_t);
template <typename T>
void launch_embed4_bwd(const T*, const long*, float*, long, int, hipStream_t);
template <typename T>
void launch_spmm_sum(const int*, const int*, const T*, T*, int, int, hipStream_t);
template <typename T>
void launch_gru_gates_fwd(const T*, const T*, const T*, T*, T*, T*, T*, long, int, hipStream_t);
template <typename T>
void launch_gru_gates_bwd(const T*, const T*, const T*, const T*, const T*, const T*, T*, T*, T*,
                          long, int, hipStream_t);
template <typename T>
void launch_attn_pool_fwd(const T*, const T*, const int*, T*, float*, int, int, hipStream_t);
template <typename T>
void launch_attn_pool_bwd(const T*, const T*, const float*, const int*, T*, T*, float*, int, int,
                          hipStream_t);
void launch_segment_max(const float*, const int*, float*, int, hipStream_t);
template <typename T>
void launch_gru_gates2_fwd(const T*, const T*, T*, T*, T*, T*, T*, long, int, hipStream_t);
template <typename T>
void launch_gru_gates2_bwd(const T*, const T*, const T*, const T*, const T*, const T*, T*, T*,
                           long, int, hipStream_t);
template <typename T>
void launch_colsum(const T*, float*, int, int, hipStream_t);
template <typename T>
void launch_spmm_sum_strided(const int*, const int*, const T*, T*, int, int, long,
                             hipStream_t);
void launch_gemm_bias2(const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
                       const __hip_bfloat16*, const __hip_bfloat16*, long,
                       const __hip_bfloat16*, long, __hip_bfloat16*, int, int, int, int,
                       hipStream_t);
void launch_gemm_bias(const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
                      const __hip_bfloat16*, const __hip_bfloat16*, __hip_bfloat16*, int, int,
                      int, int, hipStream_t);
void launch_wgrad(const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*, float*,
                  float*, int, int, int, int, hipStream_t, float* = nullptr,
                  float* = nullptr, float* = nullptr, float* = nullptr);
template <typename T>
void launch_layernorm_fwd(const T*, const float*, const float*, T*, float*, float*, long, int,
                          float, hipStream_t);
template <typename T>
void launch_layernorm_bwd(const T*, const T*, const float*, const float*, const float*, T*, long,
                          int, hipStream_t);
template <typename T>
void launch_layernorm_wgrad(const T*, const T*, const float*, const float*, float*, float*, long,
                            int, hipStream_t);
template <typename T>
void launch_bias_gelu_fwd(const T*, const float*, T*, long, int, hipStream_t);
template <typename T>
void launch_bias_gelu_bwd(const T*, const T*, const float*, T*, long, int, hipStream_t);
template <typename T>
void launch_softmax_mask_fwd(const T*, const int*, T*, T*, long, int, int, float, float,
                             unsigned long long, int, int, hipStream_t);
template <typename T>
void launch_rmsnorm_fwd(const T*, const float*, T*, float*, long, int, float, hipStream_t);
template <typename T>
void launch_rmsnorm_bwd(const T*, const T*, const float*, const float*, T*, long, int,
                        hipStream_t);
template <typename T>
void launch_rmsnorm_wgrad(const T*, const T*, const float*, float*, long, int, hipStream_t);
void launch_flash_fwd(const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
                      const int*, const float*, __hip_bfloat16*, float*, int, int, int, float,
                      int, unsigned, unsigned long long, long, long, hipStream_t);
void launch_flash_dterm(const __hip_bfloat16*, const __hip_bfloat16*, float*, int, int, int,
                        hipStream_t);
void launch_flash_dq(const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
                     const __hip_bfloat16*, const int*, const float*, const float*, const float*,
                     __hip_bfloat16*, float*, int, int, int, float, int, unsigned,
                     unsigned long long, long, long, long, hipStream_t);
void launch_wgrad2(const __hip_bfloat16*, const __hip_bfloat16*, float*, float*,
                   int, int, int, int, hipStream_t);
void launch_adamw_fused(float*, const float*, float*, float*, long, float, float, float, float,
                        float, const float*, const float*, int, void*, hipStream_t);
void launch_gemm2(const __hip_bfloat16*, const __hip_bfloat16*, const float*,
                  const __hip_bfloat16*, __hip_bfloat16*, int, int, int, hipStream_t);
void launch_flash_dkv(const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
                      const __hip_bfloat16*, const int*, const float*, const float*, const float*,
                      __hip_bfloat16*, __hip_bfloat16*, int, int, int, float, int, unsigned,
                      unsigned long long, long, long, long, hipStream_t);
template <typename T>
void launch_softmax_mask_bwd(const T*, const T*, T*, long, int, float, float,
                             unsigned long long, hipStream_t);
void dkv_prof_fetch(unsigned long long*);
void fwd_prof_fetch(unsigned long long*);
template <typename T>
void launch_dropout_add_fwd(const T*, const T*, T*, long, float, unsigned long long,
                            hipStream_t);
template <typename T>
void launch_dropout_add_bwd(const T*, T*, long, float, unsigned long long, hipStream_t);
template <typename T>
void launch_ln_res_dropout_fwd(const T*, const T*, const float*, const float*, T*, float*,
                               float*, long, int, float, float, unsigned long long, hipStream_t);
template <typename T>
void launch_ln_res_dropout_bwd(const T*, const T*, const T*, const float*, const float*,
                               const float*, T*, T*, T*, long, int, float,
                               unsigned long long, hipStream_t);
template <typename T>
void launch_ln_res_dropout_wgrad(const T*, const T*, const T*, const float*, const float*,
                                 float*, float*, long, int, float, unsigned long long,
                                 hipStream_t);
template <typename T>
void launch_embed_scatter(const T*, const long*, float*, long, int, long, hipStream_t);
void launch_relbias_wgrad(const float*, const int*, float*, long, int, int,
                          hipStream_t);
template <typename T>
void launch_relu_dropout_fwd(const T*, T*, long, float, unsigned long long, hipStream_t);
template <typename T>
void launch_relu_dropout_bwd(const T*, const T*, T*, long, float, unsigned long long,
                             hipStream_t);

#define CHECK_GPU(t) \
  TORCH_CHECK((t).is_cuda() && (t).is_contiguous(), #t " must be contiguous GPU tensor")

static hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

template <typename F>
void dispatch_float_bf16(const at::Tensor& t, const char* name, F&& f) {
  if (t.scalar_type() == at::kFloat) {
    f(float{});
  } else if (t.scalar_type() == at::kBFloat16) {
    f(__hip_bfloat16{});
  } else {
    TORCH_CHECK(false, name, ": unsupported dtype ", t.scalar_type());
  }
}

template <typename T>
const T* ptr(const at::Tensor& t) {
  return reinterpret_cast<const T*>(t.data_ptr());
}
template <typename T>
T* mptr(at::Tensor& t) {
  return reinterpret_cast<T*>(t.data_ptr());
}

at::Tensor embed4_fwd(at::Tensor tables, at::Tensor idx, bool out_bf16 = false) {
  CHECK_GPU(tables);
  CHECK_GPU(idx);
  TORCH_CHECK(tables.dim() == 3 && tables.size(0) == 4 && tables.size(2) == 32,
              "tables must be (4, V, 32)");
  TORCH_CHECK(idx.dim() == 2 && idx.size(1) == 4 && idx.scalar_type() == at::kLong);
  const int N = idx.size(0);
  const int V = tables.size(1);
  if (out_bf16 && tables.scalar_type() == at::kFloat) {
    // autocast path: gather fp32 master rows, emit bf16 directly
    auto out = at::empty({N, 128}, tables.options().dtype(at::kBFloat16));
    launch_embed4_fwd2<float, __hip_bfloat16>(
        tables.data_ptr<float>(), idx.data_ptr<long>(),
        reinterpret_cast<__hip_bfloat16*>(out.data_ptr()), N, V, cur_stream());
    return out;
  }
  auto out = at::empty({N, 128}, tables.options());
  dispatch_float_bf16(tables, "embed4_fwd", [&](auto tag) {
    using T = decltype(tag);
    launch_embed4_fwd<T>(ptr<T>(tables), idx.data_ptr<long>(), mptr<T>(out), N, V, cur_stream());
  });
  return out;
}

at::Tensor embed4_bwd(at::Tensor grad_out, at::Tensor idx, long V, long Demb,
                      c10::optional<at::Tensor> out_opt) {
  CHECK_GPU(grad_out);
  CHECK_GPU(idx);
  TORCH_CHECK(Demb == 32, "Demb must be 32");
  at::Tensor grad;
  const bool acc = out_opt.has_value();
  if (acc) {  // accumulate into the flat .grad region of the 4 tables
    grad = *out_opt;
    TORCH_CHECK(grad.is_cuda() && grad.scalar_type() == at::kFloat &&
                grad.is_contiguous() && grad.numel() == 4 * V * Demb);
  } else {
    grad = at::zeros({4, V, Demb}, grad_out.options().dtype(at::kFloat));
  }
  const long total = grad_out.numel();
  dispatch_float_bf16(grad_out, "embed4_bwd", [&](auto tag) {
    using T = decltype(tag);
    launch_embed4_bwd<T>(ptr<T>(grad_out), idx.data_ptr<long>(), grad.data_ptr<float>(), total,
                         (int)V, cur_stream());
  });
  return acc ? grad : grad.to(grad_out.scalar_type());
}

at::Tensor spmm_sum(at::Tensor indptr, at::Tensor indices, at::Tensor x) {
  CHECK_GPU(indptr);
  CHECK_GPU(indices);
  CHECK_GPU(x);
  TORCH_CHECK(indptr.scalar_type() == at::kInt && indices.scalar_type() == at::kInt);
  const int N = indptr.size(0) - 1;
  const int D = x.size(1);
  TORCH_CHECK(D % 2 == 0, "D must be even");
  TORCH_CHECK(x.size(0) == N, "x rows must match indptr");
  auto out = at::empty_like(x);
  dispatch_float_bf16(x, "spmm_sum", [&](auto tag) {
    using T = decltype(tag);
    launch_spmm_sum<T>(indptr.data_ptr<int>(), indices.data_ptr<int>(), ptr<T>(x), mptr<T>(out),
                       N, D, cur_stream());
  });
  return out;
}

std::vector<at::Tensor> gru_gates_fwd(at::Tensor gi, at::Tensor gh, at::Tensor h) {
  CHECK_GPU(gi);
  CHECK_GPU(gh);
  CHECK_GPU(h);
  const long N = h.size(0);
  const int H = h.size(1);
  TORCH_CHECK(gi.size(1) == 3 * H && gh.size(1) == 3 * H);
  TORCH_CHECK(gi.scalar_type() == h.scalar_type() && gh.scalar_type() == h.scalar_type(),
              "gi/gh/h dtypes must match");
  auto h_new = at::empty_like(h);
  auto r = at::empty_like(h);
  auto z = at::empty_like(h);
  auto n = at::empty_like(h);
  dispatch_float_bf16(h, "gru_gates_fwd", [&](auto tag) {
    using T = decltype(tag);
    launch_gru_gates_fwd<T>(ptr<T>(gi), ptr<T>(gh), ptr<T>(h), mptr<T>(h_new), mptr<T>(r),
                            mptr<T>(z), mptr<T>(n), N * H, H, cur_stream());
  });
  return {h_new, r, z, n};
}

std::vector<at::Tensor> gru_gates_bwd(at::Tensor grad_h_new, at::Tensor gh, at::Tensor h,
                                      at::Tensor r, at::Tensor z, at::Tensor n) {
  CHECK_GPU(grad_h_new);
  const long N = h.size(0);
  const int H = h.size(1);
  auto grad_gi = at::empty({N, 3 * H}, h.options());
  auto grad_gh = at::empty({N, 3 * H}, h.options());
  auto grad_h = at::empty_like(h);
  dispatch_float_bf16(h, "gru_gates_bwd", [&](auto tag) {
    using T = decltype(tag);
    launch_gru_gates_bwd<T>(ptr<T>(grad_h_new), ptr<T>(gh), ptr<T>(h), ptr<T>(r), ptr<T>(z),
                            ptr<T>(n), mptr<T>(grad_gi), mptr<T>(grad_gh), mptr<T>(grad_h), N * H,
                            H, cur_stream());
  });
  return {grad_gi, grad_gh, grad_h};
}

std::vector<at::Tensor> attn_pool_fwd(at::Tensor x, at::Tensor gate, at::Tensor node_offsets) {
  CHECK_GPU(x);
  CHECK_GPU(gate);
  CHECK_GPU(node_offsets);
  TORCH_CHECK(node_offsets.scalar_type() == at::kInt);
  TORCH_CHECK(gate.scalar_type() == x.scalar_type(), "gate/x dtypes must match");
  const int B = node_offsets.size(0) - 1;
  const int D = x.size(1);
  auto out = at::empty({B, D}, x.options());
  auto alpha = at::empty({x.size(0)}, x.options().dtype(at::kFloat));
  dispatch_float_bf16(x, "attn_pool_fwd", [&](auto tag) {
    using T = decltype(tag);
    launch_attn_pool_fwd<T>(ptr<T>(x), ptr<T>(gate), node_offsets.data_ptr<int>(), mptr<T>(out),
                            alpha.data_ptr<float>(), B, D, cur_stream());
  });
  return {out, alpha};
}

std::vector<at::Tensor> attn_pool_bwd(at::Tensor grad_out, at::Tensor x, at::Tensor alpha,
                                      at::Tensor node_offsets) {
  CHECK_GPU(grad_out);
  CHECK_GPU(x);
  CHECK_GPU(alpha);
  const int B = node_offsets.size(0) - 1;
  const int D = x.size(1);
  auto grad_x = at::empty_like(x);
  auto grad_gate = at::empty({x.size(0)}, x.options());
  auto s_ws = at::empty({x.size(0)}, x.options().dtype(at::kFloat));
  dispatch_float_bf16(x, "attn_pool_bwd", [&](auto tag) {
    using T = decltype(tag);
    launch_attn_pool_bwd<T>(ptr<T>(grad_out), ptr<T>(x), alpha.data_ptr<float>(),
                            node_offsets.data_ptr<int>(), mptr<T>(grad_x), mptr<T>(grad_gate),
                            s_ws.data_ptr<float>(), B, D, cur_stream());
  });
  return {grad_x, grad_gate};
}

at::Tensor segment_max(at::Tensor values, at::Tensor node_offsets) {
  CHECK_GPU(values);
  CHECK_GPU(node_offsets);
  TORCH_CHECK(values.scalar_type() == at::kFloat, "segment_max expects fp32");
  const int B = node_offsets.size(0) - 1;
  auto out = at::empty({B}, values.options());
  launch_segment_max(values.data_ptr<float>(), node_offsets.data_ptr<int>(),
                     out.data_ptr<float>(), B, cur_stream());
  return out;
}

// ---------------------------------------------------------------------------
// Fused GGNN: the whole n_steps message-passing loop driven from C++.
// Replaces {linear, spmm, GRUCell} x n_steps of the reference's DGL
// GatedGraphConv (ggnn.py:57-60,95) with per step:
//   wh    = gemm_bias(h, W_e^T) + b_e                      [MFMA]
//   m     = csr_segment_sum(wh)                            [spmm kernel]
//   gicat = gemm_bias([m|h], Wcat^T) + b_cat               [MFMA, split-A]
//   h     = fused_gru_gates(gicat, h)                      [elementwise]
// Wcat is the (4H, 2H) block weight matrix [W_ir|W_hr; W_iz|W_hz; W_in|0;
// 0|W_hn] so one GEMM produces all gate pre-activations.
// ---------------------------------------------------------------------------

using bf16_t = __hip_bfloat16;

static at::Tensor build_wcat(const at::Tensor& W_ih, const at::Tensor& W_hh, long H) {
  auto Wcat = at::zeros({4 * H, 2 * H}, W_ih.options());
  Wcat.narrow(0, 0, H).narrow(1, 0, H).copy_(W_ih.narrow(0, 0, H));
  Wcat.narrow(0, 0, H).narrow(1, H, H).copy_(W_hh.narrow(0, 0, H));
  Wcat.narrow(0, H, H).narrow(1, 0, H).copy_(W_ih.narrow(0, H, H));
  Wcat.narrow(0, H, H).narrow(1, H, H).copy_(W_hh.narrow(0, H, H));
  Wcat.narrow(0, 2 * H, H).narrow(1, 0, H).copy_(W_ih.narrow(0, 2 * H, H));
  Wcat.narrow(0, 3 * H, H).narrow(1, H, H).copy_(W_hh.narrow(0, 2 * H, H));
  return Wcat;
}

at::Tensor gemm_bias(at::Tensor A, at::Tensor W, c10::optional<at::Tensor> bias,
                     c10::optional<at::Tensor> addend) {
  CHECK_GPU(A);
  CHECK_GPU(W);
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 && W.scalar_type() == at::kBFloat16);
  const int N = A.size(0), K = A.size(1), COL = W.size(0);
  TORCH_CHECK(W.size(1) == K && K % 64 == 0 && COL % 128 == 0);
  auto out = at::empty({N, COL}, A.options());
  const bf16_t* b = nullptr;
  if (bias.has_value()) b = reinterpret_cast<const bf16_t*>(bias->data_ptr());
  const bf16_t* add = nullptr;
  if (addend.has_value()) {
    TORCH_CHECK(addend->is_contiguous() && addend->sizes() == out.sizes());
    add = reinterpret_cast<const bf16_t*>(addend->data_ptr());
  }
  launch_gemm_bias(ptr<bf16_t>(A), nullptr, ptr<bf16_t>(W), b, add, mptr<bf16_t>(out), N, K, K,
                   COL, cur_stream());
  return out;
}

// out(M, C) = A(K, M)^T @ B(K, C), fp32 output (wgrad shape class).
// 128-aligned shapes take the pipelined wgrad2 kernel; others the generic
// split-K kernel.
// `out` (optional): ACCUMULATE the weight grad into an existing pre-zeroed
// fp32 buffer — the flat optimizer's .grad view — via atomic epilogues.
// Skips the per-call zeros() fill and autograd's AccumulateGrad add
// (~145 fills + ~270 adds = ~2.7 ms/step on the CodeT5 step).
at::Tensor wgrad(at::Tensor A, at::Tensor B, c10::optional<at::Tensor> out_opt) {
  CHECK_GPU(A);
  CHECK_GPU(B);
  const int K = A.size(0), M = A.size(1), C = B.size(1);
  TORCH_CHECK(B.size(0) == K && M % 64 == 0 && C % 8 == 0);
  const bool acc = out_opt.has_value();
  at::Tensor out;
  if (acc) {
    out = *out_opt;
    TORCH_CHECK(out.is_cuda() && out.scalar_type() == at::kFloat &&
                out.is_contiguous() && out.numel() == (long)M * C,
                "wgrad out must be a contiguous fp32 (M, C) accumulator");
  }
  if (M % 128 == 0 && C % 128 == 0) {  // wgrad2 zero-fills K tails
    // always zero-init: whether the launcher plain-stores (one K chunk) or
    // atomically accumulates (split K) is ITS decision — duplicating the
    // chunking math here to pick empty-vs-zeros risked an uninitialized
    // accumulator if the two ever diverged (they did differ by K-floor)
    if (!acc) out = at::zeros({M, C}, A.options().dtype(at::kFloat));
    launch_wgrad2(ptr<bf16_t>(A), ptr<bf16_t>(B), out.data_ptr<float>(), nullptr,
                  K, M, C, acc ? 1 : 0, cur_stream());
    return out;
  }
  if (!acc) out = at::zeros({M, C}, A.options().dtype(at::kFloat));
  launch_wgrad(ptr<bf16_t>(A), ptr<bf16_t>(B), nullptr, out.data_ptr<float>(), nullptr,
               K, M, C, C, cur_stream());
  return out;
}

void launch_pack_gru_weights(const float*, const float*, const float*, const float*,
                             const float*, const float*, int, __hip_bfloat16*,
                             __hip_bfloat16*, __hip_bfloat16*, __hip_bfloat16*,
                             __hip_bfloat16*, __hip_bfloat16*, __hip_bfloat16*,
                             __hip_bfloat16*, hipStream_t);
void launch_gemm_gru(const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
                     const __hip_bfloat16*, const __hip_bfloat16*, __hip_bfloat16*,
                     __hip_bfloat16*, __hip_bfloat16*, __hip_bfloat16*, __hip_bfloat16*,
                     int, int, int, int, hipStream_t);

// one-launch refresh of every derived GGNN weight buffer (bf16 casts,
// Wcat/WcatT block matrix, merged bias) — see ops/flowgnn.py cache
void pack_gru_weights(at::Tensor W_e, at::Tensor b_e, at::Tensor W_ih, at::Tensor W_hh,
                      at::Tensor b_ih, at::Tensor b_hh, at::Tensor w_e16, at::Tensor b_e16,
                      at::Tensor Wcat, at::Tensor WcatT, at::Tensor b_cat, at::Tensor W_eT,
                      at::Tensor Wcat_perm, at::Tensor b_perm) {
  CHECK_GPU(W_e);
  const int H = W_e.size(0);
  TORCH_CHECK(W_e.scalar_type() == at::kFloat && W_ih.scalar_type() == at::kFloat,
              "pack_gru_weights reads fp32 master weights");
  TORCH_CHECK(Wcat.size(0) == 4 * H && Wcat.size(1) == 2 * H);
  launch_pack_gru_weights(W_e.data_ptr<float>(), b_e.data_ptr<float>(),
                          W_ih.data_ptr<float>(), W_hh.data_ptr<float>(),
                          b_ih.data_ptr<float>(), b_hh.data_ptr<float>(), H,
                          mptr<bf16_t>(w_e16), mptr<bf16_t>(b_e16), mptr<bf16_t>(Wcat),
                          mptr<bf16_t>(WcatT), mptr<bf16_t>(b_cat), mptr<bf16_t>(W_eT),
                          mptr<bf16_t>(Wcat_perm), mptr<bf16_t>(b_perm),
                          cur_stream());
}

std::vector<at::Tensor> ggnn_fused_fwd(at::Tensor indptr, at::Tensor indices, at::Tensor x,
                                       at::Tensor W_e, at::Tensor b_e, at::Tensor Wcat_perm,
                                       at::Tensor b_perm, long n_steps) {
  CHECK_GPU(x);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16, "fused GGNN path is bf16");
  const long N = x.size(0);
  const long H = x.size(1);
  TORCH_CHECK(H % 64 == 0 && (4 * H) % 128 == 0, "H must suit the MFMA tile");
  TORCH_CHECK(Wcat_perm.size(0) == 4 * H && Wcat_perm.size(1) == 2 * H &&
              b_perm.numel() == 4 * H,
              "Wcat_perm/b_perm must be prebuilt (ops/flowgnn.py cache)");
  auto stream = cur_stream();
  const long S = n_steps;
  auto opts = x.options();
  // HH[s] = hidden state entering step s; HH[S] = final output.
  auto HH = at::empty({S + 1, N, H}, opts);
  HH.select(0, 0).copy_(x);
  auto M = at::empty({S, N, H}, opts);
  auto R = at::empty({S, N, H}, opts);
  auto Z = at::empty({S, N, H}, opts);
  auto Nn = at::empty({S, N, H}, opts);
  auto HN = at::empty({S, N, H}, opts);
  auto wh = at::empty({N, H}, opts);
  const long NH = N * H;
  for (long s = 0; s < S; ++s) {
    const bf16_t* h = ptr<bf16_t>(HH) + s * NH;
    launch_gemm_bias(h, nullptr, ptr<bf16_t>(W_e), ptr<bf16_t>(b_e), nullptr, mptr<bf16_t>(wh),
                     N, H, H, H, stream);
    bf16_t* m = mptr<bf16_t>(M) + s * NH;
    launch_spmm_sum<bf16_t>(indptr.data_ptr<int>(), indices.data_ptr<int>(), ptr<bf16_t>(wh), m,
                            N, H, stream);
    // fused gate GEMM + GRU cell (gate-interleaved Wcat layout): replaces
    // the gicat GEMM + separate gru_gates2_fwd, with fp32 pre-activations
    launch_gemm_gru(m, h, ptr<bf16_t>(Wcat_perm), ptr<bf16_t>(b_perm), h,
                    mptr<bf16_t>(HH) + (s + 1) * NH, mptr<bf16_t>(R) + s * NH,
                    mptr<bf16_t>(Z) + s * NH, mptr<bf16_t>(Nn) + s * NH,
                    mptr<bf16_t>(HN) + s * NH, N, 2 * H, H, H, stream);
  }
  auto h_final = HH.select(0, S);
  return {h_final, HH, M, R, Z, Nn, HN};
}

std::vector<at::Tensor> ggnn_fused_bwd(at::Tensor grad_out, at::Tensor t_indptr,
                                       at::Tensor t_indices, at::Tensor x, at::Tensor W_eT,
                                       at::Tensor WcatT, at::Tensor HH,
                                       at::Tensor M, at::Tensor R, at::Tensor Z, at::Tensor Nn,
                                       at::Tensor HN, long n_steps,
                                       c10::optional<at::Tensor> out_we,
                                       c10::optional<at::Tensor> out_be,
                                       c10::optional<std::vector<at::Tensor>> gru_outs) {
  const long N = x.size(0);
  const long H = x.size(1);
  const long S = n_steps;
  const long NH = N * H;
  auto stream = cur_stream();
  TORCH_CHECK(WcatT.size(0) == 2 * H && WcatT.size(1) == 4 * H && W_eT.size(0) == H,
              "WcatT/W_eT must be prebuilt (ops/flowgnn.py cache)");
  auto opts = x.options();
  // per-step gate/message grads, kept for ONE batched K = S*N weight-grad
  // GEMM at the end (hipBLASLt's transpose-A kernels are the pathology the
  // custom split-K wgrad kernel replaces; see csrc/wgrad.hip header)
  auto Ggicat = at::empty({S, N, 4 * H}, opts);
  auto Gwh = at::empty({S, N, H}, opts);
  auto grad_h = grad_out.contiguous().clone();
  auto grad_hd = at::empty({N, H}, opts);
  auto grad_A = at::empty({N, 2 * H}, opts);
  for (long s = S - 1; s >= 0; --s) {
    const bf16_t* h_in = ptr<bf16_t>(HH) + s * NH;
    auto ggic = Ggicat.select(0, s);
    auto gwh = Gwh.select(0, s);
    launch_gru_gates2_bwd<bf16_t>(ptr<bf16_t>(grad_h), h_in, ptr<bf16_t>(R) + s * NH,
                                  ptr<bf16_t>(Z) + s * NH, ptr<bf16_t>(Nn) + s * NH,
                                  ptr<bf16_t>(HN) + s * NH, mptr<bf16_t>(ggic),
                                  mptr<bf16_t>(grad_hd), NH, H, stream);
    // grad_A = ggic @ Wcat : (N, 2H); left half = grad_m, right = gh-path
    launch_gemm_bias(ptr<bf16_t>(ggic), nullptr, ptr<bf16_t>(WcatT), nullptr, nullptr,
                     mptr<bf16_t>(grad_A), N, 4 * H, 4 * H, 2 * H, stream);
    // grad_m read STRIDED out of grad_A's left half (no contiguous copy)
    launch_spmm_sum_strided<bf16_t>(t_indptr.data_ptr<int>(), t_indices.data_ptr<int>(),
                                    ptr<bf16_t>(grad_A), mptr<bf16_t>(gwh), N, H,
                                    2 * H, stream);
    // grad wrt h_in = (grad_wh @ W_e) + direct z-path + gh-path, the two
    // additions fused into the GEMM epilogue as strided addends
    launch_gemm_bias2(ptr<bf16_t>(gwh), nullptr, ptr<bf16_t>(W_eT), nullptr,
                      ptr<bf16_t>(grad_hd), H, ptr<bf16_t>(grad_A) + H, 2 * H,
                      mptr<bf16_t>(grad_h), N, H, H, H, stream);
  }
  // batched weight/bias grads over all steps (K = S*N)
  auto A_g = Ggicat.view({S * N, 4 * H});
  const bool gru_direct = gru_outs.has_value();
  at::Tensor gWcat, cs4;
  if (gru_direct) {  // scatter epilogue straight into the 4 flat .grad views
    TORCH_CHECK(gru_outs->size() == 4, "gru_outs = [gW_ih, gW_hh, gb_ih, gb_hh]");
    for (auto& t : *gru_outs)
      TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kFloat && t.is_contiguous());
    TORCH_CHECK((*gru_outs)[0].numel() == 3 * H * H &&
                (*gru_outs)[1].numel() == 3 * H * H &&
                (*gru_outs)[2].numel() == 3 * H && (*gru_outs)[3].numel() == 3 * H);
  } else {
    gWcat = at::zeros({4 * H, 2 * H}, opts.dtype(at::kFloat));
    cs4 = at::zeros({4 * H}, opts.dtype(at::kFloat));
  }
  // bias grads (column sums of the gate/message grads) ride along in the
  // wgrad kernels' A-tile staging — no separate colsum pass. The gate
  // wgrad is a SPLIT-B product (no [m|h] concat) so it stays on the
  // generic split-K kernel; the W_e grad (plain A^T B, 128x128) takes the
  // tr16-subtiled wgrad2 (K tails zero-filled in staging).
  if (gru_direct)
    launch_wgrad(ptr<bf16_t>(A_g), ptr<bf16_t>(M), ptr<bf16_t>(HH), nullptr, nullptr,
                 S * N, 4 * H, 2 * H, H, stream, (*gru_outs)[0].data_ptr<float>(),
                 (*gru_outs)[1].data_ptr<float>(), (*gru_outs)[2].data_ptr<float>(),
                 (*gru_outs)[3].data_ptr<float>());
  else
    launch_wgrad(ptr<bf16_t>(A_g), ptr<bf16_t>(M), ptr<bf16_t>(HH), gWcat.data_ptr<float>(),
                 cs4.data_ptr<float>(), S * N, 4 * H, 2 * H, H, stream);
  auto A_w = Gwh.view({S * N, H});
  at::Tensor gW_e, cs_e;
  const bool acc_we = out_we.has_value();
  if (acc_we) {  // accumulate W_e/b_e grads into the flat .grad views
    gW_e = *out_we;
    cs_e = *out_be;
    TORCH_CHECK(gW_e.is_cuda() && gW_e.scalar_type() == at::kFloat &&
                gW_e.is_contiguous() && gW_e.numel() == H * H &&
                cs_e.is_cuda() && cs_e.scalar_type() == at::kFloat &&
                cs_e.numel() == H);
  } else {
    gW_e = at::zeros({H, H}, opts.dtype(at::kFloat));
    cs_e = at::zeros({H}, opts.dtype(at::kFloat));
  }
  launch_wgrad2(ptr<bf16_t>(A_w), ptr<bf16_t>(HH), gW_e.data_ptr<float>(),
                cs_e.data_ptr<float>(), S * N, H, H, acc_we ? 1 : 0, stream);
  if (gru_direct) {  // grads already accumulated in the flat views
    auto none = at::empty({0}, opts.dtype(at::kFloat));
    return {grad_h, gW_e, cs_e, none, none, none, none};
  }
  // scatter gWcat blocks back to the GRUCell weight layout (views are fine
  // as autograd outputs; no contiguous copy)
  auto gW_ih = gWcat.narrow(0, 0, 3 * H).narrow(1, 0, H);
  auto gW_hh = at::cat({gWcat.narrow(0, 0, 2 * H).narrow(1, H, H),
                        gWcat.narrow(0, 3 * H, H).narrow(1, H, H)});
  auto gb_ih = cs4.narrow(0, 0, 3 * H).contiguous();
  auto gb_hh = at::cat({cs4.narrow(0, 0, 2 * H), cs4.narrow(0, 3 * H, H)});
  return {grad_h, gW_e, cs_e, gW_ih, gW_hh, gb_ih, gb_hh};
}

std::vector<at::Tensor> gru_gates2_fwd(at::Tensor gicat, at::Tensor h) {
  CHECK_GPU(gicat);
  CHECK_GPU(h);
  const long N = h.size(0);
  const int H = h.size(1);
  auto h_new = at::empty_like(h);
  auto r = at::empty_like(h);
  auto z = at::empty_like(h);
  auto n = at::empty_like(h);
  auto hn = at::empty_like(h);
  dispatch_float_bf16(h, "gru_gates2_fwd", [&](auto tag) {
    using T = decltype(tag);
    launch_gru_gates2_fwd<T>(ptr<T>(gicat), ptr<T>(h), mptr<T>(h_new), mptr<T>(r), mptr<T>(z),
                             mptr<T>(n), mptr<T>(hn), N * H, H, cur_stream());
  });
  return {h_new, r, z, n, hn};
}

at::Tensor colsum(at::Tensor x, c10::optional<at::Tensor> out_opt) {
  CHECK_GPU(x);
  at::Tensor out;
  if (out_opt.has_value()) {  // accumulate into the flat .grad view (atomic)
    out = *out_opt;
    TORCH_CHECK(out.is_cuda() && out.scalar_type() == at::kFloat &&
                out.is_contiguous() && out.numel() == x.size(1));
  } else {
    out = at::zeros({x.size(1)}, x.options().dtype(at::kFloat));
  }
  dispatch_float_bf16(x, "colsum", [&](auto tag) {
    using T = decltype(tag);
    launch_colsum<T>(ptr<T>(x), out.data_ptr<float>(), x.size(0), x.size(1), cur_stream());
  });
  return out;
}


// ---------------------------------------------------------------------------
// Transformer kernels (csrc/transformer_kernels.hip)
// ---------------------------------------------------------------------------

std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor gamma, at::Tensor beta,
                                      double eps) {
  CHECK_GPU(x);
  TORCH_CHECK(gamma.scalar_type() == at::kFloat && beta.scalar_type() == at::kFloat);
  const int D = x.size(-1);
  const long N = x.numel() / D;
  auto y = at::empty_like(x);
  auto mean = at::empty({N}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({N}, x.options().dtype(at::kFloat));
  dispatch_float_bf16(x, "layernorm_fwd", [&](auto tag) {
    using T = decltype(tag);
    launch_layernorm_fwd<T>(ptr<T>(x), gamma.data_ptr<float>(), beta.data_ptr<float>(),
                            mptr<T>(y), mean.data_ptr<float>(), rstd.data_ptr<float>(), N, D,
                            (float)eps, cur_stream());
  });
  return {y, mean, rstd};
}

at::Tensor layernorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor gamma, at::Tensor mean,
                         at::Tensor rstd) {
  CHECK_GPU(dy);
  const int D = x.size(-1);
  const long N = x.numel() / D;
  auto dx = at::empty_like(x);
  dispatch_float_bf16(x, "layernorm_bwd", [&](auto tag) {
    using T = decltype(tag);
    launch_layernorm_bwd<T>(ptr<T>(dy), ptr<T>(x), gamma.data_ptr<float>(),
                            mean.data_ptr<float>(), rstd.data_ptr<float>(), mptr<T>(dx), N, D,
                            cur_stream());
  });
  return dx;
}

std::vector<at::Tensor> layernorm_wgrad(at::Tensor dy, at::Tensor x, at::Tensor mean,
                                        at::Tensor rstd) {
  const int D = x.size(-1);
  const long N = x.numel() / D;
  auto dgamma = at::zeros({D}, x.options().dtype(at::kFloat));
  auto dbeta = at::zeros({D}, x.options().dtype(at::kFloat));
  dispatch_float_bf16(x, "layernorm_wgrad", [&](auto tag) {
    using T = decltype(tag);
    launch_layernorm_wgrad<T>(ptr<T>(dy), ptr<T>(x), mean.data_ptr<float>(),
                              rstd.data_ptr<float>(), dgamma.data_ptr<float>(),
                              dbeta.data_ptr<float>(), N, D, cur_stream());
  });
  return {dgamma, dbeta};
}

at::Tensor bias_gelu_fwd(at::Tensor x, at::Tensor bias) {
  CHECK_GPU(x);
  TORCH_CHECK(bias.scalar_type() == at::kFloat);
  const int D = x.size(-1);
  TORCH_CHECK(D % (x.scalar_type() == at::kFloat ? 4 : 8) == 0,
              "bias_gelu: D must be a multiple of the 16-B vector width");
  auto y = at::empty_like(x);
  dispatch_float_bf16(x, "bias_gelu_fwd", [&](auto tag) {
    using T = decltype(tag);
    launch_bias_gelu_fwd<T>(ptr<T>(x), bias.data_ptr<float>(), mptr<T>(y), x.numel(), D,
                            cur_stream());
  });
  return y;
}

at::Tensor bias_gelu_bwd(at::Tensor dy, at::Tensor x, at::Tensor bias) {
  CHECK_GPU(dy);
  const int D = x.size(-1);
  auto dx = at::empty_like(x);
  dispatch_float_bf16(x, "bias_gelu_bwd", [&](auto tag) {
    using T = decltype(tag);
    launch_bias_gelu_bwd<T>(ptr<T>(dy), ptr<T>(x), bias.data_ptr<float>(), mptr<T>(dx),
                            x.numel(), D, cur_stream());
  });
  return dx;
}

static void check_softmax_geom(const at::Tensor& t, int L) {
  const int vec = t.scalar_type() == at::kBFloat16 ? 8 : 4;
  TORCH_CHECK(L % vec == 0 && L <= 64 * vec * 4, "unsupported softmax row length ", L);
}

// returns {P (pre-dropout, saved for bwd), Pd (post-dropout, feeds P@V)};
// Pd aliases P when dropout_p == 0
std::vector<at::Tensor> softmax_mask_fwd(at::Tensor S, c10::optional<at::Tensor> valid,
                                         double scale, double dropout_p, int64_t seed,
                                         bool causal) {
  CHECK_GPU(S);
  const int L = S.size(-1);
  const long R = S.numel() / L;
  check_softmax_geom(S, L);
  auto P = at::empty_like(S);
  const bool drop = dropout_p > 0.0;
  // p == 0: Pd shares storage with P via an explicit alias so autograd can
  // keep the two outputs' differentiability separate
  auto Pd = drop ? at::empty_like(S) : at::alias(P);
  const int* vptr = nullptr;
  int rows_per_batch = 1;
  if (valid.has_value()) {
    TORCH_CHECK(valid->scalar_type() == at::kInt);
    vptr = valid->data_ptr<int>();
    rows_per_batch = (int)(R / valid->size(0));
  }
  dispatch_float_bf16(S, "softmax_mask_fwd", [&](auto tag) {
    using T = decltype(tag);
    launch_softmax_mask_fwd<T>(ptr<T>(S), vptr, mptr<T>(P), drop ? mptr<T>(Pd) : nullptr, R, L,
                               rows_per_batch, (float)scale, (float)dropout_p,
                               (unsigned long long)seed, causal ? 1 : 0, (int)S.size(-2),
                               cur_stream());
  });
  return {P, Pd};
}

at::Tensor softmax_mask_bwd(at::Tensor dPd, at::Tensor P, double scale, double dropout_p,
                            int64_t seed) {
  CHECK_GPU(dPd);
  const int L = P.size(-1);
  const long R = P.numel() / L;
  check_softmax_geom(P, L);
  auto dS = at::empty_like(P);
  dispatch_float_bf16(P, "softmax_mask_bwd", [&](auto tag) {
    using T = decltype(tag);
    launch_softmax_mask_bwd<T>(ptr<T>(dPd), ptr<T>(P), mptr<T>(dS), R, L, (float)scale,
                               (float)dropout_p, (unsigned long long)seed, cur_stream());
  });
  return dS;
}


std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor gamma, double eps) {
  CHECK_GPU(x);
  TORCH_CHECK(gamma.scalar_type() == at::kFloat);
  const int D = x.size(-1);
  const long N = x.numel() / D;
  auto y = at::empty_like(x);
  auto rstd = at::empty({N}, x.options().dtype(at::kFloat));
  dispatch_float_bf16(x, "rmsnorm_fwd", [&](auto tag) {
    using T = decltype(tag);
    launch_rmsnorm_fwd<T>(ptr<T>(x), gamma.data_ptr<float>(), mptr<T>(y),
                          rstd.data_ptr<float>(), N, D, (float)eps, cur_stream());
  });
  return {y, rstd};
}

at::Tensor rmsnorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor gamma, at::Tensor rstd) {
  CHECK_GPU(dy);
  const int D = x.size(-1);
  const long N = x.numel() / D;
  auto dx = at::empty_like(x);
  dispatch_float_bf16(x, "rmsnorm_bwd", [&](auto tag) {
    using T = decltype(tag);
    launch_rmsnorm_bwd<T>(ptr<T>(dy), ptr<T>(x), gamma.data_ptr<float>(),
                          rstd.data_ptr<float>(), mptr<T>(dx), N, D, cur_stream());
  });
  return dx;
}

at::Tensor rmsnorm_wgrad(at::Tensor dy, at::Tensor x, at::Tensor rstd,
                         c10::optional<at::Tensor> out_opt) {
  const int D = x.size(-1);
  const long N = x.numel() / D;
  at::Tensor dgamma;
  if (out_opt.has_value()) {  // accumulate into the flat .grad view (atomic)
    dgamma = *out_opt;
    TORCH_CHECK(dgamma.is_cuda() && dgamma.scalar_type() == at::kFloat &&
                dgamma.is_contiguous() && dgamma.numel() == D);
  } else {
    dgamma = at::zeros({D}, x.options().dtype(at::kFloat));
  }
  dispatch_float_bf16(x, "rmsnorm_wgrad", [&](auto tag) {
    using T = decltype(tag);
    launch_rmsnorm_wgrad<T>(ptr<T>(dy), ptr<T>(x), rstd.data_ptr<float>(),
                            dgamma.data_ptr<float>(), N, D, cur_stream());
  });
  return dgamma;
}


// ---------------------------------------------------------------------------
// Flash attention (csrc/flash_attn.hip): head_dim 64, (B, L, H*64) layout
// ---------------------------------------------------------------------------

static const int* opt_valid_ptr(const c10::optional<at::Tensor>& valid) {
  if (!valid.has_value()) return nullptr;
  TORCH_CHECK(valid->scalar_type() == at::kInt && valid->is_contiguous());
  return valid->data_ptr<int>();
}

// Q/K/V may be row-strided views (slices of a fused QKV projection):
// last dim contiguous, batch stride == L * row stride
static long fa_ld(const at::Tensor& t, int L) {
  TORCH_CHECK(t.is_cuda() && t.dim() == 3 && t.stride(2) == 1 &&
                  t.stride(0) == (long)L * t.stride(1) && t.stride(1) >= t.size(2),
              "flash tensor must be a (B, L, D) row-strided view");
  return t.stride(1);
}

std::vector<at::Tensor> flash_attn_fwd(at::Tensor Q, at::Tensor K, at::Tensor V, int64_t H,
                                       c10::optional<at::Tensor> valid,
                                       c10::optional<at::Tensor> bias, double scale, bool causal,
                                       double dropout_p, int64_t seed) {
  TORCH_CHECK(Q.scalar_type() == at::kBFloat16, "flash attention is bf16");
  const int B = Q.size(0), L = Q.size(1);
  const long ldq = fa_ld(Q, L);
  const long ldkv = fa_ld(K, L);
  TORCH_CHECK(fa_ld(V, L) == ldkv, "K and V must share a row stride");
  TORCH_CHECK(Q.size(2) == H * 64, "head_dim must be 64");
  TORCH_CHECK(L % 64 == 0, "L must be a multiple of 64");
  auto O = at::empty({B, L, H * 64}, Q.options());
  auto lse = at::empty({B, H, L}, Q.options().dtype(at::kFloat));
  const float* bptr = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->scalar_type() == at::kFloat && bias->is_contiguous());
    TORCH_CHECK(bias->numel() == H * (long)L * L, "bias must be (H, L, L)");
    bptr = bias->data_ptr<float>();
  }
  launch_flash_fwd(ptr<bf16_t>(Q), ptr<bf16_t>(K), ptr<bf16_t>(V), opt_valid_ptr(valid), bptr,
                   mptr<bf16_t>(O), lse.data_ptr<float>(), B, H, L, (float)scale, causal ? 1 : 0,
                   (unsigned)(dropout_p * 256.0), (unsigned long long)seed, ldq, ldkv,
                   cur_stream());
  return {O, lse};
}

std::vector<at::Tensor> flash_attn_bwd(at::Tensor dO, at::Tensor Q, at::Tensor K, at::Tensor V,
                                       at::Tensor O, at::Tensor lse, int64_t H,
                                       c10::optional<at::Tensor> valid,
                                       c10::optional<at::Tensor> bias, double scale, bool causal,
                                       double dropout_p, int64_t seed, bool need_dbias,
                                       bool fused_grads,
                                       c10::optional<at::Tensor> dbias_accum = c10::nullopt,
                                       bool kv_fused = false) {
  CHECK_GPU(dO);
  const int B = Q.size(0), L = Q.size(1);
  const long ldq = fa_ld(Q, L);
  const long ldkv = fa_ld(K, L);
  TORCH_CHECK(fa_ld(V, L) == ldkv, "K and V must share a row stride");
  auto stream = cur_stream();
  auto Dterm = at::empty({B, H, L}, Q.options().dtype(at::kFloat));
  launch_flash_dterm(ptr<bf16_t>(dO), ptr<bf16_t>(O), Dterm.data_ptr<float>(), B, H, L, stream);
  // fused_grads: one (B, L, 3*H*64) buffer whose [q|k|v] slices the
  // kernels write directly (ld 3*H*64) — lets the attention op consume a
  // fused QKV projection with zero slice-backward scatter work
  const long HD64 = (long)H * 64;
  at::Tensor dQKV, dKV, dQ, dK, dV;
  long ld_dq = HD64, ld_dkv = HD64;
  if (fused_grads) {
    TORCH_CHECK(ldq == 3 * HD64 && ldkv == 3 * HD64,
                "fused_grads expects q/k/v slices of one (B, L, 3D) buffer");
    dQKV = at::empty({B, L, 3 * HD64}, Q.options());
    ld_dq = ld_dkv = 3 * HD64;
    dQ = dQKV.narrow(2, 0, HD64);
    dK = dQKV.narrow(2, HD64, HD64);
    dV = dQKV.narrow(2, 2 * HD64, HD64);
  } else if (kv_fused) {
    // cross-attention: dense dQ, dK/dV written as the [k|v] slices of one
    // (B, L, 2D) buffer matching a fused K/V projection of the encoder
    TORCH_CHECK(ldkv == 2 * HD64,
                "kv_fused expects k/v slices of one (B, L, 2D) buffer");
    dQ = at::empty({B, L, HD64}, Q.options());
    dKV = at::empty({B, L, 2 * HD64}, Q.options());
    ld_dkv = 2 * HD64;
    dK = dKV.narrow(2, 0, HD64);
    dV = dKV.narrow(2, HD64, HD64);
  } else {
    dQ = at::empty({B, L, HD64}, Q.options());
    dK = at::empty({B, L, HD64}, Q.options());
    dV = at::empty({B, L, HD64}, Q.options());
  }
  const float* bptr = nullptr;
  if (bias.has_value()) bptr = bias->data_ptr<float>();
  at::Tensor dBias;
  float* dbias_ptr = nullptr;
  if (dbias_accum.has_value()) {
    // shared accumulation buffer (T5: 24 layers share one position bias;
    // the dq kernel's atomics land in ONE tensor instead of 24 separate
    // dBias allocations + the autograd fan-in adds)
    TORCH_CHECK(dbias_accum->scalar_type() == at::kFloat &&
                dbias_accum->numel() == (long)H * L * L);
    dbias_ptr = dbias_accum->data_ptr<float>();
  } else if (need_dbias) {
    TORCH_CHECK(bias.has_value());
    dBias = at::zeros({H, L, L}, Q.options().dtype(at::kFloat));
    dbias_ptr = dBias.data_ptr<float>();
  }
  launch_flash_dq(ptr<bf16_t>(Q), ptr<bf16_t>(K), ptr<bf16_t>(V), ptr<bf16_t>(dO),
                  opt_valid_ptr(valid), bptr, lse.data_ptr<float>(), Dterm.data_ptr<float>(),
                  mptr<bf16_t>(dQ), dbias_ptr, B, H, L, (float)scale, causal ? 1 : 0,
                  (unsigned)(dropout_p * 256.0), (unsigned long long)seed, ldq, ldkv, ld_dq,
                  stream);
  launch_flash_dkv(ptr<bf16_t>(Q), ptr<bf16_t>(K), ptr<bf16_t>(V), ptr<bf16_t>(dO),
                   opt_valid_ptr(valid), bptr, lse.data_ptr<float>(), Dterm.data_ptr<float>(),
                   mptr<bf16_t>(dK), mptr<bf16_t>(dV), B, H, L, (float)scale,
                   causal ? 1 : 0, (unsigned)(dropout_p * 256.0), (unsigned long long)seed,
                   ldq, ldkv, ld_dkv, stream);
  if (fused_grads) {
    if (need_dbias && !dbias_accum.has_value()) return {dQKV, dBias};
    return {dQKV};
  }
  if (kv_fused) {
    if (need_dbias && !dbias_accum.has_value()) return {dQ, dKV, dBias};
    return {dQ, dKV};
  }
  if (need_dbias && !dbias_accum.has_value()) return {dQ, dK, dV, dBias};
  return {dQ, dK, dV};
}


// out(N, COL) = A(N, K) @ W(COL, K)^T [+ bias] [+ addend] — transformer
// shapes (csrc/gemm_bf16.hip). Geometry: N%128, K%64, COL%128 == 0.
at::Tensor gemm2(at::Tensor A, at::Tensor W, c10::optional<at::Tensor> bias,
                 c10::optional<at::Tensor> addend) {
  CHECK_GPU(A);
  CHECK_GPU(W);
  TORCH_CHECK(A.scalar_type() == at::kBFloat16 && W.scalar_type() == at::kBFloat16);
  const int N = A.size(0), K = A.size(1), COL = W.size(0);
  TORCH_CHECK(W.size(1) == K && N % 128 == 0 && K % 64 == 0 && COL % 128 == 0,
              "gemm2 geometry violated: ", N, "x", K, "x", COL);
  auto out = at::empty({N, COL}, A.options());
  const float* b = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->scalar_type() == at::kFloat && bias->is_contiguous());
    b = bias->data_ptr<float>();
  }
  const bf16_t* add = nullptr;
  if (addend.has_value()) add = reinterpret_cast<const bf16_t*>(addend->data_ptr());
  launch_gemm2(ptr<bf16_t>(A), ptr<bf16_t>(W), b, add, mptr<bf16_t>(out), N, K, COL,
               cur_stream());
  return out;
}


void launch_lmhead_ce_fwd(const __hip_bfloat16*, const __hip_bfloat16*, const int*,
                          float, long, int, int, int, float*, float*, float*,
                          float*, float*, int, hipStream_t);
void launch_lmhead_ce_bwd(const __hip_bfloat16*, const __hip_bfloat16*, const int*,
                          const float*, const float*, float, long, int, int, int,
                          __hip_bfloat16*, hipStream_t);

// K21 forward: per-row (loss, lse) of CE(softmax(scale * h @ Wp^T), target)
// without materializing logits. h (M,K) bf16, Wp (Vp,K) bf16 zero-padded
// past V, targets (M,) int32 with -1 = ignore. csrc/lmhead_ce.hip.
std::vector<at::Tensor> lmhead_ce_fwd(at::Tensor h, at::Tensor Wp, at::Tensor targets,
                                      double scale, int64_t V) {
  CHECK_GPU(h);
  CHECK_GPU(Wp);
  TORCH_CHECK(h.scalar_type() == at::kBFloat16 && Wp.scalar_type() == at::kBFloat16);
  TORCH_CHECK(targets.scalar_type() == at::kInt && targets.is_contiguous());
  const long M = h.size(0);
  const int K = h.size(1), Vp = Wp.size(0);
  TORCH_CHECK(Wp.size(1) == K && M % 128 == 0 && K % 64 == 0 && Vp % 128 == 0,
              "lmhead_ce geometry violated: ", M, "x", K, " vocab ", Vp);
  TORCH_CHECK(targets.numel() == M);
  const int n_chunks = (Vp / 128 + 7) / 8;
  auto fopt = h.options().dtype(at::kFloat);
  auto pmax = at::empty({M, n_chunks}, fopt);
  auto psum = at::empty({M, n_chunks}, fopt);
  auto ptgt = at::empty({M, n_chunks}, fopt);
  auto lse = at::empty({M}, fopt);
  auto loss = at::empty({M}, fopt);
  launch_lmhead_ce_fwd(ptr<bf16_t>(h), ptr<bf16_t>(Wp), targets.data_ptr<int>(),
                       (float)scale, M, K, (int)V, Vp, pmax.data_ptr<float>(),
                       psum.data_ptr<float>(), ptgt.data_ptr<float>(),
                       lse.data_ptr<float>(), loss.data_ptr<float>(), n_chunks,
                       cur_stream());
  return {loss, lse};
}

// K21 backward: dlogits (M, Vp) bf16 = (softmax - onehot(target)) * gscale,
// recomputed tile-by-tile (cols >= V zeroed).
at::Tensor lmhead_ce_bwd(at::Tensor h, at::Tensor Wp, at::Tensor targets,
                         at::Tensor lse, at::Tensor gscale, double scale, int64_t V) {
  CHECK_GPU(h);
  const long M = h.size(0);
  const int K = h.size(1), Vp = Wp.size(0);
  TORCH_CHECK(lse.scalar_type() == at::kFloat && gscale.scalar_type() == at::kFloat);
  TORCH_CHECK(lse.numel() == M && gscale.numel() == M);
  auto dlogits = at::empty({M, (long)Vp}, h.options());
  launch_lmhead_ce_bwd(ptr<bf16_t>(h), ptr<bf16_t>(Wp), targets.data_ptr<int>(),
                       lse.data_ptr<float>(), gscale.data_ptr<float>(),
                       (float)scale, M, K, (int)V, Vp,
                       mptr<bf16_t>(dlogits), cur_stream());
  return dlogits;
}

void launch_bce_logits_fwd(const float*, const float*, const float*, const float*,
                           float*, int, hipStream_t);
void launch_bce_logits_bwd(const float*, const float*, const float*, const float*,
                           const float*, const float*, float*, int, hipStream_t);

void launch_gate_pool_fwd(const __hip_bfloat16*, const __hip_bfloat16*, const float*,
                          const float*, const int*, __hip_bfloat16*, float*, int, int,
                          int, hipStream_t);
void launch_gate_pool_bwd(const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
                          const float*, const float*, const int*, __hip_bfloat16*,
                          __hip_bfloat16*, float*, float*, float*, int, int, int,
                          hipStream_t);
void launch_mlp3_fwd(const __hip_bfloat16*, const float*, const float*, const float*,
                     const float*, const float*, const float*, float*, float*, float*,
                     int, int, hipStream_t);
void launch_mlp3_bwd(const float*, const float*, const float*, const float*, const float*,
                     const float*, float*, float*, __hip_bfloat16*, int, int, hipStream_t);
void launch_mlp3_wgrad(const __hip_bfloat16*, const float*, const float*, const float*,
                       const float*, const float*, float*, float*, float*, float*,
                       float*, float*, int, int, hipStream_t);

// fused concat + gate linear + segment-softmax attention pool (flow-GNN head)
std::vector<at::Tensor> gate_pool_fwd(at::Tensor x1, at::Tensor x2, at::Tensor wg,
                                      at::Tensor bg, at::Tensor node_offsets) {
  CHECK_GPU(x1);
  CHECK_GPU(x2);
  TORCH_CHECK(x1.scalar_type() == at::kBFloat16 && x2.scalar_type() == at::kBFloat16);
  TORCH_CHECK(wg.scalar_type() == at::kFloat && bg.scalar_type() == at::kFloat);
  const int N = x1.size(0), D1 = x1.size(1), D = D1 + (int)x2.size(1);
  const int B = node_offsets.numel() - 1;
  TORCH_CHECK(x2.size(0) == N && wg.numel() == D);
  TORCH_CHECK(D == 256 && D1 == 128, "fused gate_pool is built for the "
              "128+128 concat geometry (vectorized lane mapping)");
  auto out = at::empty({B, D}, x1.options());
  auto alpha = at::empty({N}, x1.options().dtype(at::kFloat));
  launch_gate_pool_fwd(ptr<bf16_t>(x1), ptr<bf16_t>(x2), wg.data_ptr<float>(),
                       bg.data_ptr<float>(), node_offsets.data_ptr<int>(),
                       mptr<bf16_t>(out), alpha.data_ptr<float>(), B, D1, D,
                       cur_stream());
  return {out, alpha};
}

std::vector<at::Tensor> gate_pool_bwd(at::Tensor grad_out, at::Tensor x1, at::Tensor x2,
                                      at::Tensor wg, at::Tensor alpha,
                                      at::Tensor node_offsets,
                                      c10::optional<at::Tensor> out_wg,
                                      c10::optional<at::Tensor> out_bg) {
  CHECK_GPU(grad_out);
  const int N = x1.size(0), D1 = x1.size(1), D = D1 + (int)x2.size(1);
  const int B = node_offsets.numel() - 1;
  auto gx1 = at::empty_like(x1);
  auto gx2 = at::empty_like(x2);
  at::Tensor dwg, dbg;
  if (out_wg.has_value()) {  // accumulate into the flat .grad views (atomic)
    dwg = *out_wg;
    dbg = *out_bg;
    TORCH_CHECK(dwg.is_cuda() && dwg.scalar_type() == at::kFloat && dwg.numel() == D &&
                dbg.is_cuda() && dbg.scalar_type() == at::kFloat && dbg.numel() == 1);
  } else {
    dwg = at::zeros({D}, x1.options().dtype(at::kFloat));
    dbg = at::zeros({1}, x1.options().dtype(at::kFloat));
  }
  auto s_ws = at::empty({N}, x1.options().dtype(at::kFloat));
  launch_gate_pool_bwd(ptr<bf16_t>(grad_out), ptr<bf16_t>(x1), ptr<bf16_t>(x2),
                       wg.data_ptr<float>(), alpha.data_ptr<float>(),
                       node_offsets.data_ptr<int>(), mptr<bf16_t>(gx1),
                       mptr<bf16_t>(gx2), dwg.data_ptr<float>(), dbg.data_ptr<float>(),
                       s_ws.data_ptr<float>(), B, D1, D, cur_stream());
  return {gx1, gx2, dwg, dbg};
}

// fused 3-layer MLP head forward: logits + relu activations (fp32 saves)
std::vector<at::Tensor> mlp3_fwd(at::Tensor x, at::Tensor W1T, at::Tensor b1,
                                 at::Tensor W2T, at::Tensor b2, at::Tensor W3,
                                 at::Tensor b3) {
  CHECK_GPU(x);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16);
  const int B = x.size(0), D = x.size(1);
  TORCH_CHECK(D == 256 && W1T.size(0) == D && W3.numel() == D);
  auto fopt = x.options().dtype(at::kFloat);
  auto h1 = at::empty({B, D}, fopt);
  auto h2 = at::empty({B, D}, fopt);
  auto logits = at::empty({B}, fopt);
  launch_mlp3_fwd(ptr<bf16_t>(x), W1T.data_ptr<float>(), b1.data_ptr<float>(),
                  W2T.data_ptr<float>(), b2.data_ptr<float>(), W3.data_ptr<float>(),
                  b3.data_ptr<float>(), h1.data_ptr<float>(), h2.data_ptr<float>(),
                  logits.data_ptr<float>(), B, D, cur_stream());
  return {logits, h1, h2};
}

std::vector<at::Tensor> mlp3_bwd(at::Tensor dlogits, at::Tensor x, at::Tensor h1,
                                 at::Tensor h2, at::Tensor W1, at::Tensor W2,
                                 at::Tensor W3,
                                 c10::optional<std::vector<at::Tensor>> outs) {
  CHECK_GPU(dlogits);
  const int B = x.size(0), D = x.size(1);
  auto fopt = x.options().dtype(at::kFloat);
  auto dh1 = at::empty({B, D}, fopt);
  auto dh2 = at::empty({B, D}, fopt);
  auto dx = at::empty_like(x);
  launch_mlp3_bwd(dlogits.data_ptr<float>(), h1.data_ptr<float>(), h2.data_ptr<float>(),
                  W1.data_ptr<float>(), W2.data_ptr<float>(), W3.data_ptr<float>(),
                  dh1.data_ptr<float>(), dh2.data_ptr<float>(), mptr<bf16_t>(dx), B, D,
                  cur_stream());
  at::Tensor dW1, dW2, dW3, db1, db2, db3;
  if (outs.has_value()) {  // accumulate into the flat .grad views
    TORCH_CHECK(outs->size() == 6, "outs = [dW1, db1, dW2, db2, dW3, db3]");
    dW1 = (*outs)[0]; db1 = (*outs)[1]; dW2 = (*outs)[2];
    db2 = (*outs)[3]; dW3 = (*outs)[4]; db3 = (*outs)[5];
    for (auto& t : *outs)
      TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kFloat && t.is_contiguous());
    TORCH_CHECK(dW1.numel() == (long)D * D && dW2.numel() == (long)D * D &&
                dW3.numel() == D && db1.numel() == D && db2.numel() == D &&
                db3.numel() == 1);
  } else {
    // the wgrad tail accumulates (+=) so every output starts zeroed
    dW1 = at::zeros({D, D}, fopt);
    dW2 = at::zeros({D, D}, fopt);
    dW3 = at::zeros({1, D}, fopt);
    db1 = at::zeros({D}, fopt);
    db2 = at::zeros({D}, fopt);
    db3 = at::zeros({1}, fopt);
  }
  launch_mlp3_wgrad(ptr<bf16_t>(x), h1.data_ptr<float>(), h2.data_ptr<float>(),
                    dh1.data_ptr<float>(), dh2.data_ptr<float>(),
                    dlogits.data_ptr<float>(), dW1.data_ptr<float>(),
                    dW2.data_ptr<float>(), dW3.data_ptr<float>(), db1.data_ptr<float>(),
                    db2.data_ptr<float>(), db3.data_ptr<float>(), B, D, cur_stream());
  if (outs.has_value()) return {dx};
  return {dx, dW1, dW2, dW3, db1, db2, db3};
}

void adamw_fused(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v, double lr,
                 double beta1, double beta2, double eps, double weight_decay, at::Tensor step,
                 c10::optional<at::Tensor> gclip, bool l2_mode,
                 c10::optional<at::Tensor> p16) {
  CHECK_GPU(p);
  TORCH_CHECK(p.scalar_type() == at::kFloat && g.scalar_type() == at::kFloat);
  TORCH_CHECK(p.numel() == g.numel() && p.numel() == m.numel() && p.numel() == v.numel());
  TORCH_CHECK(step.is_cuda() && step.scalar_type() == at::kFloat && step.numel() == 1,
              "step must be a device float[1] step counter (pre-incremented)");
  const float* gc = nullptr;
  if (gclip.has_value()) {
    TORCH_CHECK(gclip->is_cuda() && gclip->scalar_type() == at::kFloat);
    gc = gclip->data_ptr<float>();
  }
  void* s16 = nullptr;
  if (p16.has_value()) {
    TORCH_CHECK(p16->is_cuda() && p16->scalar_type() == at::kBFloat16 &&
                p16->numel() == p.numel() && p16->is_contiguous(),
                "p16 must be a contiguous bf16 shadow of p");
    s16 = p16->data_ptr();
  }
  launch_adamw_fused(p.data_ptr<float>(), g.data_ptr<float>(), m.data_ptr<float>(),
                     v.data_ptr<float>(), p.numel(), (float)lr, (float)beta1, (float)beta2,
                     (float)eps, (float)weight_decay, step.data_ptr<float>(), gc,
                     l2_mode ? 1 : 0, s16, cur_stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "deepdfa_amd MI355X (gfx950) kernels";
  m.def("embed4_fwd", &embed4_fwd, pybind11::arg("tables"), pybind11::arg("idx"), pybind11::arg("out_bf16") = false);
  m.def("embed4_bwd", &embed4_bwd, pybind11::arg("grad_out"), pybind11::arg("idx"),
        pybind11::arg("V"), pybind11::arg("Demb"),
        pybind11::arg("out") = pybind11::none());
  m.def("spmm_sum", &spmm_sum);
  m.def("gru_gates_fwd", &gru_gates_fwd);
  m.def("gru_gates_bwd", &gru_gates_bwd);
  m.def("attn_pool_fwd", &attn_pool_fwd);
  m.def("attn_pool_bwd", &attn_pool_bwd);
  m.def("segment_max", &segment_max);
  m.def("gemm_bias", &gemm_bias);
  m.def("wgrad", &wgrad, pybind11::arg("A"), pybind11::arg("B"),
        pybind11::arg("out") = pybind11::none());
  m.def("gru_gates2_fwd", &gru_gates2_fwd);
  m.def("colsum", &colsum, pybind11::arg("x"), pybind11::arg("out") = pybind11::none());
  m.def("ggnn_fused_fwd", &ggnn_fused_fwd);
  m.def("pack_gru_weights", &pack_gru_weights);
  m.def("bce_logits_fwd", [](at::Tensor logits, at::Tensor labels,
                             c10::optional<at::Tensor> weight,
                             c10::optional<at::Tensor> pos_weight) {
    CHECK_GPU(logits);
    auto out2 = at::empty({2}, logits.options());
    launch_bce_logits_fwd(
        logits.data_ptr<float>(), labels.data_ptr<float>(),
        weight.has_value() ? weight->data_ptr<float>() : nullptr,
        pos_weight.has_value() ? pos_weight->data_ptr<float>() : nullptr,
        out2.data_ptr<float>(), logits.numel(), cur_stream());
    return out2;
  });
  m.def("bce_logits_bwd", [](at::Tensor logits, at::Tensor labels,
                             c10::optional<at::Tensor> weight,
                             c10::optional<at::Tensor> pos_weight,
                             at::Tensor grad, at::Tensor out2) {
    auto dlogits = at::empty_like(logits);
    launch_bce_logits_bwd(
        logits.data_ptr<float>(), labels.data_ptr<float>(),
        weight.has_value() ? weight->data_ptr<float>() : nullptr,
        pos_weight.has_value() ? pos_weight->data_ptr<float>() : nullptr,
        grad.data_ptr<float>(), out2.data_ptr<float>(),
        dlogits.data_ptr<float>(), logits.numel(), cur_stream());
    return dlogits;
  });
  m.def("gate_pool_fwd", &gate_pool_fwd);
  m.def("gate_pool_bwd", &gate_pool_bwd, pybind11::arg("grad_out"), pybind11::arg("x1"),
        pybind11::arg("x2"), pybind11::arg("wg"), pybind11::arg("alpha"),
        pybind11::arg("node_offsets"), pybind11::arg("out_wg") = pybind11::none(),
        pybind11::arg("out_bg") = pybind11::none());
  m.def("mlp3_fwd", &mlp3_fwd);
  m.def("mlp3_bwd", &mlp3_bwd, pybind11::arg("dlogits"), pybind11::arg("x"),
        pybind11::arg("h1"), pybind11::arg("h2"), pybind11::arg("W1"),
        pybind11::arg("W2"), pybind11::arg("W3"),
        pybind11::arg("outs") = pybind11::none());
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("layernorm_wgrad", &layernorm_wgrad);
  m.def("bias_gelu_fwd", &bias_gelu_fwd);
  m.def("bias_gelu_bwd", &bias_gelu_bwd);
  m.def("softmax_mask_fwd", &softmax_mask_fwd);
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("rmsnorm_wgrad", &rmsnorm_wgrad, pybind11::arg("dy"), pybind11::arg("x"),
        pybind11::arg("rstd"), pybind11::arg("out") = pybind11::none());
  m.def("flash_attn_fwd", &flash_attn_fwd);
  m.def("flash_attn_bwd", &flash_attn_bwd, pybind11::arg("dO"), pybind11::arg("Q"),
        pybind11::arg("K"), pybind11::arg("V"), pybind11::arg("O"), pybind11::arg("lse"),
        pybind11::arg("H"), pybind11::arg("valid"), pybind11::arg("bias"),
        pybind11::arg("scale"), pybind11::arg("causal"), pybind11::arg("dropout_p"),
        pybind11::arg("seed"), pybind11::arg("need_dbias"), pybind11::arg("fused_grads"),
        pybind11::arg("dbias_accum") = pybind11::none(),
        pybind11::arg("kv_fused") = false);
  m.def("lmhead_ce_fwd", &lmhead_ce_fwd);
  m.def("lmhead_ce_bwd", &lmhead_ce_bwd);
  m.def("relbias_wgrad", [](at::Tensor dbias, at::Tensor buckets, int64_t nb) {
    CHECK_GPU(dbias);
    TORCH_CHECK(dbias.scalar_type() == at::kFloat &&
                buckets.scalar_type() == at::kInt && buckets.is_contiguous());
    const int H = dbias.size(0);
    const long QK = dbias.numel() / H;
    TORCH_CHECK(buckets.numel() == QK);
    auto dw = at::zeros({nb, (long)H}, dbias.options());
    launch_relbias_wgrad(dbias.data_ptr<float>(), buckets.data_ptr<int>(),
                         dw.data_ptr<float>(), QK, H, (int)nb, cur_stream());
    return dw;
  });
  m.def("relu_dropout_fwd", [](at::Tensor x, double p, int64_t seed) {
    CHECK_GPU(x);
    auto out = at::empty_like(x);
    dispatch_float_bf16(x, "relu_dropout_fwd", [&](auto tag) {
      using T = decltype(tag);
      launch_relu_dropout_fwd<T>(ptr<T>(x), mptr<T>(out), x.numel(), (float)p,
                                 (unsigned long long)seed, cur_stream());
    });
    return out;
  });
  m.def("relu_dropout_bwd", [](at::Tensor dy, at::Tensor x, double p, int64_t seed) {
    CHECK_GPU(dy);
    auto dx = at::empty_like(dy);
    dispatch_float_bf16(dy, "relu_dropout_bwd", [&](auto tag) {
      using T = decltype(tag);
      launch_relu_dropout_bwd<T>(ptr<T>(dy), ptr<T>(x), mptr<T>(dx), dy.numel(),
                                 (float)p, (unsigned long long)seed, cur_stream());
    });
    return dx;
  });
  m.def("dropout_add_fwd", [](at::Tensor h, at::Tensor res, double p, int64_t seed) {
    CHECK_GPU(h);
    CHECK_GPU(res);
    TORCH_CHECK(h.numel() == res.numel());
    TORCH_CHECK(h.numel() % (h.scalar_type() == at::kFloat ? 4 : 8) == 0,
                "dropout_add: numel must be a multiple of the 16-B vector");
    auto out = at::empty_like(h);
    dispatch_float_bf16(h, "dropout_add_fwd", [&](auto tag) {
      using T = decltype(tag);
      launch_dropout_add_fwd<T>(ptr<T>(h), ptr<T>(res), mptr<T>(out), h.numel(), (float)p,
                                (unsigned long long)seed, cur_stream());
    });
    return out;
  });
  m.def("dropout_add_bwd", [](at::Tensor dy, double p, int64_t seed) {
    CHECK_GPU(dy);
    auto dh = at::empty_like(dy);
    dispatch_float_bf16(dy, "dropout_add_bwd", [&](auto tag) {
      using T = decltype(tag);
      launch_dropout_add_bwd<T>(ptr<T>(dy), mptr<T>(dh), dy.numel(), (float)p,
                                (unsigned long long)seed, cur_stream());
    });
    return dh;
  });
  m.def("ln_res_dropout_fwd", [](at::Tensor h, at::Tensor res, at::Tensor gamma,
                                 at::Tensor beta, double eps, double p, int64_t seed) {
    CHECK_GPU(h);
    CHECK_GPU(res);
    const int D = h.size(-1);
    TORCH_CHECK(D % 256 == 0, "ln_res_dropout: D must be a multiple of 256");
    const long N = h.numel() / D;
    auto y = at::empty_like(h);
    auto mean = at::empty({N}, h.options().dtype(at::kFloat));
    auto rstd = at::empty({N}, h.options().dtype(at::kFloat));
    dispatch_float_bf16(h, "ln_res_dropout_fwd", [&](auto tag) {
      using T = decltype(tag);
      launch_ln_res_dropout_fwd<T>(ptr<T>(h), ptr<T>(res), gamma.data_ptr<float>(),
                                   beta.data_ptr<float>(), mptr<T>(y),
                                   mean.data_ptr<float>(), rstd.data_ptr<float>(), N, D,
                                   (float)eps, (float)p, (unsigned long long)seed,
                                   cur_stream());
    });
    return std::vector<at::Tensor>{y, mean, rstd};
  });
  m.def("ln_res_dropout_bwd", [](at::Tensor dy, at::Tensor h, at::Tensor res,
                                 at::Tensor gamma, at::Tensor mean, at::Tensor rstd,
                                 double p, int64_t seed) {
    CHECK_GPU(dy);
    const int D = dy.size(-1);
    const long N = dy.numel() / D;
    auto dz = at::empty_like(dy);
    auto dh = at::empty_like(dy);
    auto z = at::empty_like(dy);
    dispatch_float_bf16(dy, "ln_res_dropout_bwd", [&](auto tag) {
      using T = decltype(tag);
      launch_ln_res_dropout_bwd<T>(ptr<T>(dy), ptr<T>(h), ptr<T>(res),
                                   gamma.data_ptr<float>(), mean.data_ptr<float>(),
                                   rstd.data_ptr<float>(), mptr<T>(dz), mptr<T>(dh),
                                   mptr<T>(z), N, D, (float)p, (unsigned long long)seed,
                                   cur_stream());
    });
    return std::vector<at::Tensor>{dz, dh, z};
  });
  m.def("ln_res_dropout_wgrad", [](at::Tensor dy, at::Tensor h, at::Tensor res,
                                   at::Tensor mean, at::Tensor rstd, double p,
                                   int64_t seed) {
    CHECK_GPU(dy);
    const int D = dy.size(-1);
    const long N = dy.numel() / D;
    auto dgamma = at::zeros({(long)D}, dy.options().dtype(at::kFloat));
    auto dbeta = at::zeros({(long)D}, dy.options().dtype(at::kFloat));
    dispatch_float_bf16(dy, "ln_res_dropout_wgrad", [&](auto tag) {
      using T = decltype(tag);
      launch_ln_res_dropout_wgrad<T>(ptr<T>(dy), ptr<T>(h), ptr<T>(res),
                                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                                     dgamma.data_ptr<float>(), dbeta.data_ptr<float>(), N,
                                     D, (float)p, (unsigned long long)seed, cur_stream());
    });
    return std::vector<at::Tensor>{dgamma, dbeta};
  });
  m.def("embed_scatter", [](at::Tensor dY, at::Tensor idx, long num_rows, long padding_idx,
                            c10::optional<at::Tensor> out_opt) {
    CHECK_GPU(dY);
    TORCH_CHECK(idx.scalar_type() == at::kLong && idx.is_contiguous());
    const int D = dY.size(-1);
    const long N = dY.numel() / D;
    at::Tensor dW;
    if (out_opt.has_value()) {  // accumulate into the flat .grad view (atomic)
      dW = *out_opt;
      TORCH_CHECK(dW.is_cuda() && dW.scalar_type() == at::kFloat &&
                  dW.is_contiguous() && dW.numel() == num_rows * (long)D);
    } else {
      dW = at::zeros({num_rows, (long)D}, dY.options().dtype(at::kFloat));
    }
    dispatch_float_bf16(dY, "embed_scatter", [&](auto tag) {
      using T = decltype(tag);
      launch_embed_scatter<T>(ptr<T>(dY), idx.data_ptr<long>(),
                              dW.data_ptr<float>(), N, D, padding_idx,
                              cur_stream());
    });
    return dW;
  }, pybind11::arg("dY"), pybind11::arg("idx"), pybind11::arg("num_rows"),
     pybind11::arg("padding_idx"), pybind11::arg("out") = pybind11::none());
  m.def("fwd_prof", []() {
    auto t = torch::zeros({8}, torch::dtype(torch::kLong));
    fwd_prof_fetch(reinterpret_cast<unsigned long long*>(t.data_ptr<long>()));
    return t;
  });
  m.def("dkv_prof", []() {
    // variant-9 instrumentation readout: per-segment cycle totals
    auto t = torch::zeros({8}, torch::dtype(torch::kLong));
    dkv_prof_fetch(reinterpret_cast<unsigned long long*>(t.data_ptr<long>()));
    return t;
  });
  m.def("gemm2", &gemm2);
  m.def("adamw_fused", &adamw_fused, pybind11::arg("p"), pybind11::arg("g"),
        pybind11::arg("m"), pybind11::arg("v"), pybind11::arg("lr"),
        pybind11::arg("beta1"), pybind11::arg("beta2"), pybind11::arg("eps"),
        pybind11::arg("weight_decay"), pybind11::arg("step"),
        pybind11::arg("gclip") = pybind11::none(),
        pybind11::arg("l2_mode") = false,
        pybind11::arg("p16") = pybind11::none());
  m.def("softmax_mask_bwd", &softmax_mask_bwd);
  m.def("ggnn_fused_bwd", &ggnn_fused_bwd, pybind11::arg("grad_out"),
        pybind11::arg("t_indptr"), pybind11::arg("t_indices"), pybind11::arg("x"),
        pybind11::arg("W_eT"), pybind11::arg("WcatT"), pybind11::arg("HH"),
        pybind11::arg("M"), pybind11::arg("R"), pybind11::arg("Z"), pybind11::arg("Nn"),
        pybind11::arg("HN"), pybind11::arg("n_steps"),
        pybind11::arg("out_we") = pybind11::none(),
        pybind11::arg("out_be") = pybind11::none(),
        pybind11::arg("gru_outs") = pybind11::none());
}
