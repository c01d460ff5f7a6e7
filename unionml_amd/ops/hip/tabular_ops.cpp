// Torch bindings for the CDNA4 tabular hot-path kernels
// (kernels: tabular_kernels.hip). All launches go to the current torch
// stream so they compose with torch.cuda.graphs capture and RCCL work.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

extern "C" {
void launch_standardize_fit(const float*, long long, int, float*, float*, float,
                            double*, hipStream_t);
void launch_standardize_apply(const float*, long long, int, int, const float*,
                              const float*, unsigned short*, hipStream_t);
// generalized-geometry kernels (tabular_gen.hip)
int gen_rt_for_hid(int);
int launch_mlp_step_gen(const unsigned short*, const int*, int, int, int, int,
                        int, const unsigned short*, const float*, float*, int,
                        int, float, hipStream_t);
void launch_reduce_adam_gen(const float*, int, int, int, int, int, float*,
                            unsigned short*, float*, float*, int*, float*,
                            float, float, float, float, unsigned short*, float*,
                            unsigned*, hipStream_t);
int launch_mlp_predict_gen(const float*, int, int, int, int, int, const float*,
                           const float*, const unsigned short*, const float*,
                           int*, float*, hipStream_t);
void launch_adam_step_gen(float*, unsigned short*, const float*, float*, float*,
                          int*, int, int, int, int, float, float, float, float,
                          unsigned short*, hipStream_t);
void launch_mlp_step(const unsigned short*, const int*, int, const unsigned short*,
                     const unsigned short*, const float*, float*, float,
                     hipStream_t);
int launch_mlp_train_steps(const unsigned short*, const int*, long long, int, int,
                           float*, unsigned short*, float*, float*, int*, float*,
                           float, float, float, float, hipStream_t);
int launch_mlp_step_fused(const unsigned short*, const int*, int,
                          const unsigned short*, const unsigned short*, float*,
                          unsigned short*, float*, float*, int*, float*, unsigned*,
                          float*, float, float, float, float, float, int, float*,
                          unsigned short*, hipStream_t);
void launch_mlp_predict(const float*, int, const float*, const float*,
                        const unsigned short*, const unsigned short*, const float*,
                        int*, float*, hipStream_t);
void launch_adam_step(float*, unsigned short*, const float*, float*, float*, int*,
                      float, float, float, float, unsigned short*, hipStream_t);
}

namespace {

hipStream_t current_stream() {
  return at::hip::getCurrentHIPStream().stream();
}

void check(const torch::Tensor& t, torch::ScalarType dtype, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be a device tensor");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  TORCH_CHECK(t.scalar_type() == dtype, name, " has wrong dtype");
}

const unsigned short* bf16_ptr(const torch::Tensor& t) {
  return reinterpret_cast<const unsigned short*>(t.data_ptr());
}

unsigned short* bf16_mut_ptr(torch::Tensor& t) {
  return reinterpret_cast<unsigned short*>(t.data_ptr());
}

}  // namespace

void standardize_fit(torch::Tensor X, torch::Tensor mean, torch::Tensor invstd,
                     double eps) {
  check(X, torch::kFloat32, "X");
  check(mean, torch::kFloat32, "mean");
  check(invstd, torch::kFloat32, "invstd");
  const int D = (int)X.size(1);
  TORCH_CHECK(mean.numel() == D && invstd.numel() == D, "mean/invstd size mismatch");
  double* scratch_ptr = nullptr;
  torch::Tensor scratch;
  if (D <= 256 && 256 % D == 0) {
    scratch = torch::zeros({2 * D}, mean.options().dtype(torch::kFloat64));
    scratch_ptr = scratch.data_ptr<double>();
  }
  launch_standardize_fit(X.data_ptr<float>(), X.size(0), D, mean.data_ptr<float>(),
                         invstd.data_ptr<float>(), (float)eps, scratch_ptr,
                         current_stream());
}

void standardize_apply(torch::Tensor X, torch::Tensor mean, torch::Tensor invstd,
                       torch::Tensor out_bf16) {
  check(X, torch::kFloat32, "X");
  check(out_bf16, torch::kBFloat16, "out");
  TORCH_CHECK(out_bf16.dim() == 2 && out_bf16.size(0) == X.size(0) &&
                  out_bf16.size(1) >= X.size(1),
              "out must be [N][Dout] with Dout >= D");
  launch_standardize_apply(X.data_ptr<float>(), X.size(0), (int)X.size(1),
                           (int)out_bf16.size(1), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), bf16_mut_ptr(out_bf16),
                           current_stream());
}

void mlp_step(torch::Tensor Xbf, torch::Tensor y, torch::Tensor W1bf,
              torch::Tensor W2bf, torch::Tensor master, torch::Tensor grads,
              double invBtot) {
  check(Xbf, torch::kBFloat16, "Xbf");
  check(y, torch::kInt32, "y");
  check(W1bf, torch::kBFloat16, "W1bf");
  check(W2bf, torch::kBFloat16, "W2bf");
  check(master, torch::kFloat32, "master");
  check(grads, torch::kFloat32, "grads");
  TORCH_CHECK(Xbf.size(1) == 64, "IN must be 64");
  TORCH_CHECK(W1bf.numel() == 64 * 32 && W2bf.numel() == 32 * 16, "weight shapes");
  TORCH_CHECK(grads.numel() >= 2609, "grads must hold 2608 params + loss");
  launch_mlp_step(bf16_ptr(Xbf), y.data_ptr<int>(), (int)Xbf.size(0),
                  bf16_ptr(W1bf), bf16_ptr(W2bf), master.data_ptr<float>(),
                  grads.data_ptr<float>(), (float)invBtot, current_stream());
}

bool mlp_step_fused(torch::Tensor Xbf, torch::Tensor y, torch::Tensor W1bf,
                    torch::Tensor W2bf, torch::Tensor master, torch::Tensor bfmirror,
                    torch::Tensor m, torch::Tensor v, torch::Tensor t_dev,
                    torch::Tensor slabs, torch::Tensor counter,
                    torch::Tensor loss_out, double invBtot, double lr,
                    double beta1, double beta2, double eps,
                    c10::optional<torch::Tensor> grads_out = c10::nullopt,
                    c10::optional<torch::Tensor> wimg = c10::nullopt) {
  check(Xbf, torch::kBFloat16, "Xbf");
  check(y, torch::kInt32, "y");
  check(master, torch::kFloat32, "master");
  check(bfmirror, torch::kBFloat16, "bfmirror");
  check(slabs, torch::kFloat32, "slabs");
  check(counter, torch::kUInt32, "counter");
  check(loss_out, torch::kFloat32, "loss_out");
  TORCH_CHECK(Xbf.size(1) == 64, "IN must be 64");
  TORCH_CHECK(slabs.dim() == 2 && slabs.size(1) == 2624, "slabs must be [n][2624]");
  float* grads_ptr = nullptr;
  if (grads_out.has_value()) {
    check(*grads_out, torch::kFloat32, "grads_out");
    TORCH_CHECK(grads_out->numel() >= 2609, "grads_out must hold params + loss");
    grads_ptr = grads_out->data_ptr<float>();
  }
  unsigned short* wimg_ptr = nullptr;
  if (wimg.has_value()) {
    check(*wimg, torch::kBFloat16, "wimg");
    TORCH_CHECK(wimg->numel() == 4224, "wimg must be the packed 4224-elem image buffer");
    wimg_ptr = bf16_mut_ptr(*wimg);
  }
  const int rc = launch_mlp_step_fused(
      bf16_ptr(Xbf), y.data_ptr<int>(), (int)Xbf.size(0), bf16_ptr(W1bf),
      bf16_ptr(W2bf), master.data_ptr<float>(), bf16_mut_ptr(bfmirror),
      m.data_ptr<float>(), v.data_ptr<float>(), t_dev.data_ptr<int>(),
      slabs.data_ptr<float>(), (unsigned*)counter.data_ptr(),
      loss_out.data_ptr<float>(), (float)invBtot, (float)lr, (float)beta1,
      (float)beta2, (float)eps, (int)slabs.size(0), grads_ptr, wimg_ptr,
      current_stream());
  TORCH_CHECK(rc != -2, "mlp_step_fused: hipFuncSetAttribute(LDS) failed");
  return rc == 0;
}

bool mlp_train_steps(torch::Tensor Xbf, torch::Tensor y, int64_t batch,
                     int64_t n_steps, torch::Tensor master, torch::Tensor bfmirror,
                     torch::Tensor m, torch::Tensor v, torch::Tensor t_dev,
                     torch::Tensor loss_out, double lr, double beta1, double beta2,
                     double eps) {
  check(Xbf, torch::kBFloat16, "Xbf");
  check(y, torch::kInt32, "y");
  check(master, torch::kFloat32, "master");
  check(bfmirror, torch::kBFloat16, "bfmirror");
  check(m, torch::kFloat32, "m");
  check(v, torch::kFloat32, "v");
  check(t_dev, torch::kInt32, "t_dev");
  check(loss_out, torch::kFloat32, "loss_out");
  TORCH_CHECK(Xbf.size(1) == 64, "IN must be 64");
  const int rc = launch_mlp_train_steps(
      bf16_ptr(Xbf), y.data_ptr<int>(), Xbf.size(0), (int)batch, (int)n_steps,
      master.data_ptr<float>(), bf16_mut_ptr(bfmirror), m.data_ptr<float>(),
      v.data_ptr<float>(), t_dev.data_ptr<int>(), loss_out.data_ptr<float>(),
      (float)lr, (float)beta1, (float)beta2, (float)eps, current_stream());
  TORCH_CHECK(rc != -2, "mlp_train_steps: hipFuncSetAttribute(LDS) failed");
  return rc == 0;   // false -> shape constraints unmet, caller falls back
}

void mlp_predict(torch::Tensor X, torch::Tensor mean, torch::Tensor invstd,
                 torch::Tensor W1bf, torch::Tensor W2bf, torch::Tensor master,
                 torch::Tensor preds, c10::optional<torch::Tensor> probs) {
  check(X, torch::kFloat32, "X");
  check(preds, torch::kInt32, "preds");
  TORCH_CHECK(X.size(1) == 64, "IN must be 64");
  float* probs_ptr = nullptr;
  if (probs.has_value()) {
    check(*probs, torch::kFloat32, "probs");
    probs_ptr = probs->data_ptr<float>();
  }
  launch_mlp_predict(X.data_ptr<float>(), (int)X.size(0), mean.data_ptr<float>(),
                     invstd.data_ptr<float>(), bf16_ptr(W1bf), bf16_ptr(W2bf),
                     master.data_ptr<float>(), preds.data_ptr<int>(), probs_ptr,
                     current_stream());
}

void adam_step(torch::Tensor master, torch::Tensor bfmirror, torch::Tensor grads,
               torch::Tensor m, torch::Tensor v, torch::Tensor t_dev, double lr,
               double beta1, double beta2, double eps,
               c10::optional<torch::Tensor> wimg = c10::nullopt) {
  check(master, torch::kFloat32, "master");
  check(bfmirror, torch::kBFloat16, "bfmirror");
  check(grads, torch::kFloat32, "grads");
  check(m, torch::kFloat32, "m");
  check(v, torch::kFloat32, "v");
  check(t_dev, torch::kInt32, "t_dev");
  unsigned short* wimg_ptr = nullptr;
  if (wimg.has_value()) {
    check(*wimg, torch::kBFloat16, "wimg");
    TORCH_CHECK(wimg->numel() == 4224, "wimg must be the packed 4224-elem image buffer");
    wimg_ptr = bf16_mut_ptr(*wimg);
  }
  launch_adam_step(master.data_ptr<float>(), bf16_mut_ptr(bfmirror),
                   grads.data_ptr<float>(), m.data_ptr<float>(), v.data_ptr<float>(),
                   t_dev.data_ptr<int>(), (float)lr, (float)beta1, (float)beta2,
                   (float)eps, wimg_ptr, current_stream());
}

// ---------------------------------------------------------------------------
// generalized-geometry (any in_features/hidden/classes) entry points
// ---------------------------------------------------------------------------

bool mlp_step_gen(torch::Tensor Xbf, torch::Tensor y, int64_t hid, int64_t cls,
                  int64_t rt, torch::Tensor wimg, torch::Tensor master,
                  torch::Tensor slabs, double invBtot) {
  check(Xbf, torch::kBFloat16, "Xbf");
  check(y, torch::kInt32, "y");
  check(wimg, torch::kBFloat16, "wimg");
  check(master, torch::kFloat32, "master");
  check(slabs, torch::kFloat32, "slabs");
  const int inp = (int)Xbf.size(1);
  const int cpad = cls <= 16 ? 16 : 32;
  const int nparam = (int)(inp * hid + hid + hid * cpad + cpad);
  TORCH_CHECK(inp % 32 == 0, "staged input width must be a multiple of 32");
  TORCH_CHECK(master.numel() >= nparam, "master too small for geometry");
  TORCH_CHECK(wimg.numel() >= (int64_t)hid * inp + hid * 32 + cpad * hid,
              "wimg too small for geometry");
  TORCH_CHECK(slabs.dim() == 2 && slabs.size(1) >= nparam + 2,
              "slabs must be [n][>= nparam+2]");
  const int rc = launch_mlp_step_gen(
      bf16_ptr(Xbf), y.data_ptr<int>(), (int)Xbf.size(0), inp, (int)hid,
      (int)cls, (int)rt, bf16_ptr(wimg), master.data_ptr<float>(),
      slabs.data_ptr<float>(), (int)slabs.size(1), (int)slabs.size(0),
      (float)invBtot, current_stream());
  TORCH_CHECK(rc != -2, "mlp_step_gen: hipFuncSetAttribute(LDS) failed");
  TORCH_CHECK(rc != -3, "mlp_step_gen: unsupported geometry (hid=", hid,
              " inp=", inp, " cls=", cls, " rt=", rt, ")");
  return rc == 0;  // false -> more WGs than slab rows, caller re-sizes
}

void reduce_adam_gen(torch::Tensor slabs, int64_t n_wg, int64_t inp, int64_t hid,
                     int64_t cpad, torch::Tensor master, torch::Tensor bfmirror,
                     torch::Tensor m, torch::Tensor v, torch::Tensor t_dev,
                     torch::Tensor counter, torch::Tensor loss_out, double lr,
                     double beta1, double beta2, double eps,
                     c10::optional<torch::Tensor> wimg = c10::nullopt,
                     c10::optional<torch::Tensor> grads_out = c10::nullopt) {
  check(counter, torch::kUInt32, "counter");
  check(slabs, torch::kFloat32, "slabs");
  check(master, torch::kFloat32, "master");
  check(bfmirror, torch::kBFloat16, "bfmirror");
  check(loss_out, torch::kFloat32, "loss_out");
  TORCH_CHECK(cpad == 16 || cpad == 32, "cpad must be 16 or 32");
  const int nparam = (int)(inp * hid + hid + hid * cpad + cpad);
  TORCH_CHECK(slabs.dim() == 2 && slabs.size(1) >= nparam + 2, "slab stride");
  TORCH_CHECK(n_wg >= 1 && n_wg <= slabs.size(0), "n_wg out of range");
  unsigned short* wimg_ptr = nullptr;
  if (wimg.has_value()) {
    check(*wimg, torch::kBFloat16, "wimg");
    wimg_ptr = bf16_mut_ptr(*wimg);
  }
  float* grads_ptr = nullptr;
  if (grads_out.has_value()) {
    check(*grads_out, torch::kFloat32, "grads_out");
    TORCH_CHECK(grads_out->numel() >= nparam + 1, "grads_out too small");
    grads_ptr = grads_out->data_ptr<float>();
  }
  launch_reduce_adam_gen(slabs.data_ptr<float>(), (int)n_wg,
                         (int)slabs.size(1), (int)inp, (int)hid, (int)cpad,
                         master.data_ptr<float>(), bf16_mut_ptr(bfmirror),
                         m.data_ptr<float>(), v.data_ptr<float>(),
                         t_dev.data_ptr<int>(), loss_out.data_ptr<float>(),
                         (float)lr, (float)beta1, (float)beta2, (float)eps,
                         wimg_ptr, grads_ptr, (unsigned*)counter.data_ptr(),
                         current_stream());
}

void mlp_predict_gen(torch::Tensor X, int64_t inp, int64_t hid, int64_t cls,
                     torch::Tensor mean, torch::Tensor invstd, torch::Tensor wimg,
                     torch::Tensor master, torch::Tensor preds,
                     c10::optional<torch::Tensor> probs) {
  check(X, torch::kFloat32, "X");
  check(wimg, torch::kBFloat16, "wimg");
  check(master, torch::kFloat32, "master");
  check(preds, torch::kInt32, "preds");
  TORCH_CHECK(mean.numel() == X.size(1) && invstd.numel() == X.size(1),
              "mean/invstd must match raw feature width");
  float* probs_ptr = nullptr;
  if (probs.has_value()) {
    check(*probs, torch::kFloat32, "probs");
    probs_ptr = probs->data_ptr<float>();
  }
  const int rc = launch_mlp_predict_gen(
      X.data_ptr<float>(), (int)X.size(0), (int)X.size(1), (int)inp, (int)hid,
      (int)cls, mean.data_ptr<float>(), invstd.data_ptr<float>(), bf16_ptr(wimg),
      master.data_ptr<float>(), preds.data_ptr<int>(), probs_ptr,
      current_stream());
  TORCH_CHECK(rc == 0, "mlp_predict_gen: unsupported geometry or LDS failure");
}

void adam_step_gen(torch::Tensor master, torch::Tensor bfmirror,
                   torch::Tensor grads, torch::Tensor m, torch::Tensor v,
                   torch::Tensor t_dev, int64_t inp, int64_t hid, int64_t cpad,
                   double lr, double beta1, double beta2, double eps,
                   c10::optional<torch::Tensor> wimg = c10::nullopt) {
  check(master, torch::kFloat32, "master");
  check(bfmirror, torch::kBFloat16, "bfmirror");
  check(grads, torch::kFloat32, "grads");
  TORCH_CHECK(cpad == 16 || cpad == 32, "cpad must be 16 or 32");
  const int nparam = (int)(inp * hid + hid + hid * cpad + cpad);
  TORCH_CHECK(master.numel() >= nparam, "master too small for geometry");
  unsigned short* wimg_ptr = nullptr;
  if (wimg.has_value()) {
    check(*wimg, torch::kBFloat16, "wimg");
    wimg_ptr = bf16_mut_ptr(*wimg);
  }
  launch_adam_step_gen(master.data_ptr<float>(), bf16_mut_ptr(bfmirror),
                       grads.data_ptr<float>(), m.data_ptr<float>(),
                       v.data_ptr<float>(), t_dev.data_ptr<int>(), nparam,
                       (int)inp, (int)hid, (int)cpad, (float)lr, (float)beta1,
                       (float)beta2, (float)eps, wimg_ptr, current_stream());
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("standardize_fit", &standardize_fit, "column mean/invstd (CDNA4)");
  m.def("standardize_apply", &standardize_apply, "(x-mean)*invstd -> bf16 (CDNA4)");
  m.def("mlp_step", &mlp_step, "fused MLP fwd+bwd step (CDNA4 MFMA)");
  m.def("mlp_step_fused", &mlp_step_fused,
        "fully-fused step: fwd+bwd + cross-WG slab reduction + Adam in one "
        "launch; with grads_out, reduce-only (the DP pre-collective kernel)",
        py::arg("Xbf"), py::arg("y"), py::arg("W1bf"), py::arg("W2bf"),
        py::arg("master"), py::arg("bfmirror"), py::arg("m"), py::arg("v"),
        py::arg("t_dev"), py::arg("slabs"), py::arg("counter"),
        py::arg("loss_out"), py::arg("invBtot"), py::arg("lr"), py::arg("beta1"),
        py::arg("beta2"), py::arg("eps"), py::arg("grads_out") = c10::nullopt,
        py::arg("wimg") = c10::nullopt);
  m.def("mlp_train_steps", &mlp_train_steps,
        "persistent multi-step training kernel (weights+Adam resident in LDS)");
  m.def("mlp_predict", &mlp_predict, "fused standardize+fwd+argmax (CDNA4 MFMA)");
  m.def("adam_step", &adam_step, "fused Adam on flat master params (CDNA4)",
        py::arg("master"), py::arg("bfmirror"), py::arg("grads"), py::arg("m"),
        py::arg("v"), py::arg("t_dev"), py::arg("lr"), py::arg("beta1"),
        py::arg("beta2"), py::arg("eps"), py::arg("wimg") = c10::nullopt);
  m.def("gen_rt_for_hid", &gen_rt_for_hid,
        "rows-per-workgroup for a supported hidden width (0 = unsupported)");
  m.def("mlp_step_gen", &mlp_step_gen,
        "generalized fused step: any (in,hid,cls) geometry; double-buffered "
        "K-tiled MFMA fwd+bwd producing per-WG gradient slabs",
        py::arg("Xbf"), py::arg("y"), py::arg("hid"), py::arg("cls"),
        py::arg("rt"), py::arg("wimg"), py::arg("master"), py::arg("slabs"),
        py::arg("invBtot"));
  m.def("reduce_adam_gen", &reduce_adam_gen,
        "wide-grid slab reduction + fused Adam (grads_out: reduce-only, the "
        "DP pre-collective mode)",
        py::arg("slabs"), py::arg("n_wg"), py::arg("inp"), py::arg("hid"),
        py::arg("cpad"), py::arg("master"), py::arg("bfmirror"), py::arg("m"),
        py::arg("v"), py::arg("t_dev"), py::arg("counter"), py::arg("loss_out"),
        py::arg("lr"), py::arg("beta1"), py::arg("beta2"), py::arg("eps"),
        py::arg("wimg") = c10::nullopt, py::arg("grads_out") = c10::nullopt);
  m.def("mlp_predict_gen", &mlp_predict_gen,
        "generalized fused standardize+fwd+argmax",
        py::arg("X"), py::arg("inp"), py::arg("hid"), py::arg("cls"),
        py::arg("mean"), py::arg("invstd"), py::arg("wimg"), py::arg("master"),
        py::arg("preds"), py::arg("probs") = c10::nullopt);
  m.def("adam_step_gen", &adam_step_gen,
        "generalized fused Adam (runtime param count; DP path)",
        py::arg("master"), py::arg("bfmirror"), py::arg("grads"), py::arg("m"),
        py::arg("v"), py::arg("t_dev"), py::arg("inp"), py::arg("hid"),
        py::arg("cpad"), py::arg("lr"), py::arg("beta1"), py::arg("beta2"),
        py::arg("eps"), py::arg("wimg") = c10::nullopt);
}
