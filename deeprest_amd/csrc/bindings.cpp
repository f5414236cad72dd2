// Python bindings for the deeprest_amd CDNA4 kernels (torch extension).
//
// Tensor checking / allocation / stream plumbing lives here; the kernels in
// *.hip are torch-free and exposed through a C ABI.
#include <torch/extension.h>
#include <ATen/cuda/CUDAContext.h>
#include <c10/cuda/CUDAGuard.h>
#include <hip/hip_runtime.h>

#include <vector>

// ---- C ABI launchers from the .hip translation units ----
extern "C" {
void dr_layernorm_fwd(const void* x, const float* w, const float* b, void* y,
                      float* mean, float* rstd, int64_t n_rows, int D, float eps,
                      int is_bf16, hipStream_t stream);
void dr_layernorm_bwd(const void* dy, const void* x, const float* w,
                      const float* mean, const float* rstd, void* dx,
                      float* dwdb_part, int n_blocks, int64_t n_rows, int D,
                      int is_bf16, hipStream_t stream);
void dr_pinball_fwd(const void* out, const float* labels, const float* quantiles,
                    int Q, int64_t N, float inv_count, float* loss, int is_bf16,
                    hipStream_t stream);
void dr_pinball_bwd(const void* out, const float* labels, const float* quantiles,
                    int Q, int64_t N, const float* grad_loss, float inv_n,
                    void* dout, int is_bf16, hipStream_t stream);
void dr_fused_adam(const int64_t* meta, int nt, int64_t total, float lr, float beta1,
                   float beta2, float eps, float weight_decay, int step,
                   hipStream_t stream);
void dr_fused_adam_dev(const int64_t* meta, int nt, int64_t total,
                       const float* lr_ptr, float beta1, float beta2, float eps,
                       float weight_decay, const int* step_ptr,
                       hipStream_t stream);
void dr_gru_fwd(const void* xg, const void* gamma, const void* beta,
                const void* w_hh, const float* b_hh, const void* h0, void* h_all,
                void* saves, int B, int TT, int C, int reverse, int save,
                int fp8, int is_bf16, hipStream_t stream);
void dr_gru_bwd(const void* grad_h, const void* w_img, const void* w_fwd,
                const void* xg, const void* gamma, const void* beta,
                const float* b_hh, const void* h0, const void* h_all,
                const void* saves, void* dpre, float* dh0, int B, int TT, int C,
                int reverse, int is_bf16, hipStream_t stream);
void dr_gru_bwd_reduce(const void* dpre, const void* gamma, const void* xg,
                       void* dxg, float* dgamma, float* dbeta, void* gamma_pi,
                       void* xg_pi, int64_t BT, int C, int is_bf16,
                       hipStream_t stream);
void dr_mha_fwd(const void* q, const void* k, const void* v, void* o, float* lse,
                int64_t BH, int T_len, int D, float scale, int is_bf16,
                hipStream_t stream);
void dr_mha_bwd(const void* q, const void* k, const void* v, const void* o,
                const void* dout, const float* lse, float* dq, void* dk, void* dv,
                int64_t BH, int T_len, int D, float scale, int is_bf16,
                hipStream_t stream);
}

namespace {

bool is_bf16(const at::Tensor& t) { return t.scalar_type() == at::kBFloat16; }

void check_dtype(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.scalar_type() == at::kBFloat16 || t.scalar_type() == at::kFloat,
              name, " must be bf16 or f32, got ", t.scalar_type());
}

hipStream_t cur_stream() {
  return at::cuda::getCurrentCUDAStream().stream();
}

// ------------------------------------------------------------- layer norm
std::vector<at::Tensor> layer_norm_forward(at::Tensor x, at::Tensor w, at::Tensor b,
                                           double eps) {
  const at::cuda::CUDAGuard guard(x.device());
  check_dtype(x, "x");
  TORCH_CHECK(x.is_contiguous() && w.is_contiguous() && b.is_contiguous());
  TORCH_CHECK(w.scalar_type() == at::kFloat && b.scalar_type() == at::kFloat,
              "layer_norm weight/bias must be f32");
  int D = (int)x.size(-1);
  int64_t n_rows = x.numel() / D;
  TORCH_CHECK(w.numel() == D && b.numel() == D);
  auto y = at::empty_like(x);
  auto mean = at::empty({n_rows}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({n_rows}, x.options().dtype(at::kFloat));
  dr_layernorm_fwd(x.data_ptr(), w.data_ptr<float>(), b.data_ptr<float>(),
                   y.data_ptr(), mean.data_ptr<float>(), rstd.data_ptr<float>(),
                   n_rows, D, (float)eps, is_bf16(x), cur_stream());
  return {y, mean, rstd};
}

std::vector<at::Tensor> layer_norm_backward(at::Tensor dy, at::Tensor x, at::Tensor w,
                                            at::Tensor mean, at::Tensor rstd) {
  const at::cuda::CUDAGuard guard(x.device());
  TORCH_CHECK(dy.is_contiguous() && x.is_contiguous());
  TORCH_CHECK(dy.scalar_type() == x.scalar_type());
  int D = (int)x.size(-1);
  int64_t n_rows = x.numel() / D;
  int n_blocks = (int)std::min<int64_t>((n_rows + 3) / 4, 1024);
  if (n_blocks == 0) n_blocks = 1;
  auto dx = at::empty_like(x);
  auto part = at::zeros({n_blocks, 2, D}, x.options().dtype(at::kFloat));
  dr_layernorm_bwd(dy.data_ptr(), x.data_ptr(), w.data_ptr<float>(),
                   mean.data_ptr<float>(), rstd.data_ptr<float>(), dx.data_ptr(),
                   part.data_ptr<float>(), n_blocks, n_rows, D, is_bf16(x),
                   cur_stream());
  auto sums = part.sum(0);  // (2, D)
  return {dx, sums[0], sums[1]};
}

// ---------------------------------------------------------------- pinball
at::Tensor pinball_forward(at::Tensor outputs, at::Tensor labels, at::Tensor q) {
  const at::cuda::CUDAGuard guard(outputs.device());
  check_dtype(outputs, "pinball outputs");
  TORCH_CHECK(labels.scalar_type() == at::kFloat, "pinball labels must be f32");
  TORCH_CHECK(outputs.is_contiguous() && labels.is_contiguous());
  int Q = (int)outputs.size(-1);
  TORCH_CHECK(Q <= 8, "at most 8 quantiles");
  int64_t N = outputs.numel() / Q;
  TORCH_CHECK(labels.numel() == N);
  auto loss = at::zeros({}, outputs.options().dtype(at::kFloat));
  float inv_count = N > 0 ? 1.0f / (float)N : 0.f;
  dr_pinball_fwd(outputs.data_ptr(), labels.data_ptr<float>(),
                 q.data_ptr<float>(), Q, N, inv_count, loss.data_ptr<float>(),
                 is_bf16(outputs), cur_stream());
  return loss;
}

at::Tensor pinball_backward(at::Tensor grad, at::Tensor outputs, at::Tensor labels,
                            at::Tensor q) {
  const at::cuda::CUDAGuard guard(outputs.device());
  int Q = (int)outputs.size(-1);
  int64_t N = outputs.numel() / Q;
  auto dout = at::empty_like(outputs);
  // keep the upstream grad on-device (no .item() host sync — the kernel reads
  // it, so the whole loss backward is hipGraph-capturable)
  TORCH_CHECK(grad.is_cuda() && grad.scalar_type() == at::kFloat && grad.numel() == 1,
              "pinball grad must be a scalar f32 CUDA tensor");
  float inv_n = N > 0 ? 1.f / (float)N : 0.f;
  auto gc = grad.contiguous();
  dr_pinball_bwd(outputs.data_ptr(), labels.data_ptr<float>(),
                 q.data_ptr<float>(), Q, N, gc.data_ptr<float>(), inv_n,
                 dout.data_ptr(), is_bf16(outputs), cur_stream());
  return dout;
}

// ------------------------------------------------------------- fused adam
void fused_adam(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                int64_t step, double lr, double beta1, double beta2, double eps,
                double weight_decay) {
  TORCH_CHECK(!params.empty());
  const at::cuda::CUDAGuard guard(params[0].device());
  int nt = (int)params.size();
  std::vector<int64_t> meta(5 * nt);
  int64_t total = 0;
  for (int i = 0; i < nt; ++i) {
    TORCH_CHECK(params[i].scalar_type() == at::kFloat, "fused_adam expects f32 params");
    TORCH_CHECK(params[i].is_contiguous() && grads[i].is_contiguous());
    meta[i] = (int64_t)params[i].data_ptr();
    meta[nt + i] = (int64_t)grads[i].data_ptr();
    meta[2 * nt + i] = (int64_t)ms[i].data_ptr();
    meta[3 * nt + i] = (int64_t)vs[i].data_ptr();
    total += params[i].numel();
    meta[4 * nt + i] = total;  // cumulative end offsets
  }
  auto meta_t = at::from_blob(meta.data(), {(int64_t)meta.size()},
                              at::TensorOptions().dtype(at::kLong))
                    .to(params[0].device(), /*non_blocking=*/false);
  dr_fused_adam(meta_t.data_ptr<int64_t>(), nt, total, (float)lr, (float)beta1,
                (float)beta2, (float)eps, (float)weight_decay, (int)step,
                cur_stream());
}

// hipGraph-capturable variant: the pointer table is prebuilt on device (ops/
// adam.py caches it while pointers are stable) and the step counter is a
// device int32 scalar, so a replayed graph keeps exact bias correction with
// zero host work per step.
void fused_adam_capturable(at::Tensor meta_dev, int64_t nt, int64_t total,
                           at::Tensor step_t, at::Tensor lr_t, double beta1,
                           double beta2, double eps, double weight_decay) {
  const at::cuda::CUDAGuard guard(meta_dev.device());
  TORCH_CHECK(meta_dev.is_cuda() && meta_dev.scalar_type() == at::kLong &&
              meta_dev.is_contiguous());
  TORCH_CHECK(step_t.is_cuda() && step_t.scalar_type() == at::kInt &&
              step_t.numel() == 1);
  TORCH_CHECK(lr_t.is_cuda() && lr_t.scalar_type() == at::kFloat &&
              lr_t.numel() == 1);
  dr_fused_adam_dev(meta_dev.data_ptr<int64_t>(), (int)nt, total,
                    lr_t.data_ptr<float>(), (float)beta1, (float)beta2,
                    (float)eps, (float)weight_decay, step_t.data_ptr<int>(),
                    cur_stream());
}

// -------------------------------------------------------------------- gru
std::vector<at::Tensor> gru_seq_forward(at::Tensor xg, at::Tensor w_hh,
                                        at::Tensor b_hh, at::Tensor h0,
                                        at::Tensor gamma, at::Tensor beta,
                                        bool reverse, bool save, bool fp8) {
  TORCH_CHECK(!(fp8 && save), "fp8 GRU path is inference-only (no saves)");
  const at::cuda::CUDAGuard guard(xg.device());
  check_dtype(xg, "x_gates");
  TORCH_CHECK(xg.dim() == 3, "x_gates must be (B, T, 3H)");
  TORCH_CHECK(h0.dim() == 3, "h0 must be (B, C, H)");
  int B = (int)xg.size(0), TT = (int)xg.size(1);
  int C = (int)h0.size(1), H = (int)h0.size(2);
  TORCH_CHECK(H == 128, "fused GRU kernel requires hidden size 128, got ", H);
  TORCH_CHECK(C >= 3, "fused GRU kernel requires >= 3 components, got ", C);
  TORCH_CHECK(xg.size(2) == 3 * H && w_hh.size(0) == 3 * H && w_hh.size(1) == H);
  TORCH_CHECK(gamma.sizes() == at::IntArrayRef({C, 3 * H}));
  TORCH_CHECK(b_hh.scalar_type() == at::kFloat, "b_hh must be f32");
  auto dt = xg.scalar_type();
  TORCH_CHECK(w_hh.scalar_type() == dt && h0.scalar_type() == dt &&
                  gamma.scalar_type() == dt && beta.scalar_type() == dt,
              "gru operand dtypes must match x_gates");
  TORCH_CHECK(xg.is_contiguous() && w_hh.is_contiguous() && h0.is_contiguous() &&
              gamma.is_contiguous() && beta.is_contiguous() && b_hh.is_contiguous());

  TORCH_CHECK(xg.numel() < (1LL << 31),
              "x_gates too large for 32-bit staging offsets");
  auto h_all = at::empty({B, TT, C, H}, xg.options());
  auto saves = save ? at::empty({B, TT, C, 2 * H}, xg.options())
                    : at::empty({0}, xg.options());
  // the GEMM streams a bf16 weight image from L2 regardless of T
  auto w_gemm = w_hh.scalar_type() == at::kBFloat16
                    ? w_hh
                    : w_hh.to(at::kBFloat16);
  dr_gru_fwd(xg.data_ptr(), gamma.data_ptr(), beta.data_ptr(), w_gemm.data_ptr(),
             b_hh.data_ptr<float>(), h0.data_ptr(), h_all.data_ptr(),
             save ? saves.data_ptr() : nullptr, B, TT, C, reverse ? 1 : 0,
             save ? 1 : 0, fp8 ? 1 : 0, dt == at::kBFloat16, cur_stream());
  return {h_all, saves};
}

std::vector<at::Tensor> gru_seq_backward_kernel(at::Tensor grad_h, at::Tensor w_img,
                                                at::Tensor w_fwd, at::Tensor xg,
                                                at::Tensor gamma, at::Tensor beta,
                                                at::Tensor b_hh, at::Tensor h0,
                                                at::Tensor h_all, at::Tensor saves,
                                                bool reverse) {
  const at::cuda::CUDAGuard guard(grad_h.device());
  int B = (int)grad_h.size(0), TT = (int)grad_h.size(1);
  int C = (int)grad_h.size(2), H = (int)grad_h.size(3);
  TORCH_CHECK(H == 128);
  TORCH_CHECK(saves.numel() == (int64_t)B * TT * C * 2 * H,
              "gru backward: saves tensor missing or wrong size (forward must "
              "run with save=true)");
  TORCH_CHECK(w_img.scalar_type() == at::kBFloat16 &&
                  w_img.sizes() == at::IntArrayRef({H, 3 * H}) &&
                  w_img.is_contiguous(),
              "w_img must be the (H, 3H) bf16 pi-permuted W image");
  TORCH_CHECK(w_fwd.scalar_type() == at::kBFloat16 &&
                  w_fwd.sizes() == at::IntArrayRef({3 * H, H}) &&
                  w_fwd.is_contiguous(),
              "w_fwd must be the (3H, H) bf16 weight image");
  TORCH_CHECK(b_hh.scalar_type() == at::kFloat && b_hh.is_contiguous());
  auto dt = grad_h.scalar_type();
  TORCH_CHECK(xg.scalar_type() == dt && gamma.scalar_type() == dt &&
              beta.scalar_type() == dt);
  TORCH_CHECK(grad_h.is_contiguous() && h_all.is_contiguous() && saves.is_contiguous());
  auto dpre = at::empty({B, TT, C, 4 * H}, grad_h.options());
  auto dh0 = at::empty({B, C, H}, grad_h.options().dtype(at::kFloat));
  dr_gru_bwd(grad_h.data_ptr(), w_img.data_ptr(), w_fwd.data_ptr(), xg.data_ptr(),
             gamma.data_ptr(), beta.data_ptr(), b_hh.data_ptr<float>(),
             h0.data_ptr(), h_all.data_ptr(), saves.data_ptr(), dpre.data_ptr(),
             dh0.data_ptr<float>(), B, TT, C, reverse ? 1 : 0,
             dt == at::kBFloat16, cur_stream());
  return {dpre, dh0};
}

// single-pass reductions over dpre: dxg, dgamma, dbeta
std::vector<at::Tensor> gru_bwd_reduce(at::Tensor dpre, at::Tensor gamma,
                                       at::Tensor xg) {
  const at::cuda::CUDAGuard guard(dpre.device());
  TORCH_CHECK(dpre.dim() == 4 && dpre.size(3) == 512);
  int64_t BT = dpre.size(0) * dpre.size(1);
  int C = (int)dpre.size(2);
  TORCH_CHECK(xg.is_contiguous() && gamma.is_contiguous() && dpre.is_contiguous());
  auto dxg = at::empty({dpre.size(0), dpre.size(1), 384}, dpre.options());
  auto dgamma = at::zeros({C, 384}, dpre.options().dtype(at::kFloat));
  auto dbeta = at::zeros({C, 512}, dpre.options().dtype(at::kFloat));
  // pi-layout staging images (see pi_permute_rows_kernel)
  auto gamma_pi = at::empty_like(gamma);
  auto xg_pi = at::empty_like(xg);
  dr_gru_bwd_reduce(dpre.data_ptr(), gamma.data_ptr(), xg.data_ptr(),
                    dxg.data_ptr(), dgamma.data_ptr<float>(),
                    dbeta.data_ptr<float>(), gamma_pi.data_ptr(),
                    xg_pi.data_ptr(), BT, C,
                    dpre.scalar_type() == at::kBFloat16, cur_stream());
  return {dxg, dgamma, dbeta};
}

// -------------------------------------------------------------------- mha
std::vector<at::Tensor> mha_forward(at::Tensor q, at::Tensor k, at::Tensor v,
                                    double scale) {
  const at::cuda::CUDAGuard guard(q.device());
  check_dtype(q, "q");
  TORCH_CHECK(q.dim() == 4, "q must be (B, H, T, D)");
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  int64_t B = q.size(0), NH = q.size(1);
  int T_len = (int)q.size(2), D = (int)q.size(3);
  TORCH_CHECK(D >= 8 && D <= 64 && D % 8 == 0, "head dim must be 8..64, mult of 8");
  auto o = at::empty_like(q);
  auto lse = at::empty({B, NH, T_len}, q.options().dtype(at::kFloat));
  dr_mha_fwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
             lse.data_ptr<float>(), B * NH, T_len, D, (float)scale,
             is_bf16(q), cur_stream());
  return {o, lse};
}

std::vector<at::Tensor> mha_backward(at::Tensor q, at::Tensor k, at::Tensor v,
                                     at::Tensor o, at::Tensor dout, at::Tensor lse,
                                     double scale) {
  const at::cuda::CUDAGuard guard(q.device());
  int64_t B = q.size(0), NH = q.size(1);
  int T_len = (int)q.size(2), D = (int)q.size(3);
  TORCH_CHECK(dout.is_contiguous() && o.is_contiguous());
  auto dq = at::zeros(q.sizes(), q.options().dtype(at::kFloat));
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  dr_mha_bwd(q.data_ptr(), k.data_ptr(), v.data_ptr(), o.data_ptr(),
             dout.data_ptr(), lse.data_ptr<float>(), dq.data_ptr<float>(),
             dk.data_ptr(), dv.data_ptr(), B * NH, T_len, D, (float)scale,
             is_bf16(q), cur_stream());
  return {dq, dk, dv};
}

}  // namespace

void register_featurize(py::module_& m);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  register_featurize(m);
  m.def("layer_norm_forward", &layer_norm_forward);
  m.def("layer_norm_backward", &layer_norm_backward);
  m.def("pinball_forward", &pinball_forward);
  m.def("pinball_backward", &pinball_backward);
  m.def("fused_adam", &fused_adam);
  m.def("fused_adam_capturable", &fused_adam_capturable);
  m.def("gru_seq_forward", &gru_seq_forward, py::arg("xg"), py::arg("w_hh"),
        py::arg("b_hh"), py::arg("h0"), py::arg("gamma"), py::arg("beta"),
        py::arg("reverse"), py::arg("save"), py::arg("fp8") = false);
  m.def("gru_seq_backward_kernel", &gru_seq_backward_kernel);
  m.def("gru_bwd_reduce", &gru_bwd_reduce);
  m.def("mha_forward", &mha_forward);
  m.def("mha_backward", &mha_backward);
}
