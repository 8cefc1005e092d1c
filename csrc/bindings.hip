// Python bindings for the MI355X flow-GNN kernels (deepdfa_amd._C).

#include <torch/extension.h>

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <c10/hip/HIPStream.h>

// the launchers are defined (at global scope) in flowgnn_kernels.hip
template <typename T>
void launch_embed4_fwd(const T*, const long*, T*, int, int, hipStream_t);
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

at::Tensor embed4_fwd(at::Tensor tables, at::Tensor idx) {
  CHECK_GPU(tables);
  CHECK_GPU(idx);
  TORCH_CHECK(tables.dim() == 3 && tables.size(0) == 4 && tables.size(2) == 32,
              "tables must be (4, V, 32)");
  TORCH_CHECK(idx.dim() == 2 && idx.size(1) == 4 && idx.scalar_type() == at::kLong);
  const int N = idx.size(0);
  const int V = tables.size(1);
  auto out = at::empty({N, 128}, tables.options());
  dispatch_float_bf16(tables, "embed4_fwd", [&](auto tag) {
    using T = decltype(tag);
    launch_embed4_fwd<T>(ptr<T>(tables), idx.data_ptr<long>(), mptr<T>(out), N, V, cur_stream());
  });
  return out;
}

at::Tensor embed4_bwd(at::Tensor grad_out, at::Tensor idx, long V, long Demb) {
  CHECK_GPU(grad_out);
  CHECK_GPU(idx);
  TORCH_CHECK(Demb == 32, "Demb must be 32");
  auto grad = at::zeros({4, V, Demb}, grad_out.options().dtype(at::kFloat));
  const long total = grad_out.numel();
  dispatch_float_bf16(grad_out, "embed4_bwd", [&](auto tag) {
    using T = decltype(tag);
    launch_embed4_bwd<T>(ptr<T>(grad_out), idx.data_ptr<long>(), grad.data_ptr<float>(), total,
                         (int)V, cur_stream());
  });
  return grad.to(grad_out.scalar_type());
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

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "deepdfa_amd MI355X (gfx950) kernels";
  m.def("embed4_fwd", &embed4_fwd);
  m.def("embed4_bwd", &embed4_bwd);
  m.def("spmm_sum", &spmm_sum);
  m.def("gru_gates_fwd", &gru_gates_fwd);
  m.def("gru_gates_bwd", &gru_gates_bwd);
  m.def("attn_pool_fwd", &attn_pool_fwd);
  m.def("attn_pool_bwd", &attn_pool_bwd);
  m.def("segment_max", &segment_max);
}
