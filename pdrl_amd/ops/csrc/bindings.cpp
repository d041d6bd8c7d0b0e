// Python bindings for the pdrl_amd CDNA4 HIP kernels.
#include <torch/extension.h>

#include <vector>

std::vector<at::Tensor> seq_lstm_forward_hip(
    const at::Tensor&, const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&);
std::vector<at::Tensor> seq_lstm_backward_core_hip(
    const at::Tensor&, const c10::optional<at::Tensor>&,
    const c10::optional<at::Tensor>&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&);
at::Tensor gae_hip(const at::Tensor&, double, double, const at::Tensor&);
std::vector<at::Tensor> vtrace_hip(const at::Tensor&, const at::Tensor&,
                                   const at::Tensor&, const at::Tensor&,
                                   const at::Tensor&, double, double, double,
                                   double);
void l2norm_sq_hip(const at::Tensor&, at::Tensor&);
void rmsprop_step_hip(at::Tensor&, const at::Tensor&, at::Tensor&,
                      const at::Tensor&, double, double, double, double);
void adam_step_hip(at::Tensor&, const at::Tensor&, at::Tensor&, at::Tensor&,
                   at::Tensor&, const at::Tensor&, double, double, double,
                   double, double);
void soft_update_hip(const std::vector<at::Tensor>&,
                     const std::vector<at::Tensor>&, double);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("seq_lstm_forward", &seq_lstm_forward_hip,
        "fused body+LSTM+heads forward (gfx950)");
  m.def("seq_lstm_backward_core", &seq_lstm_backward_core_hip,
        "fused BPTT backward core (gfx950)");
  m.def("gae", &gae_hip, "GAE reverse scan");
  m.def("vtrace", &vtrace_hip, "fused V-trace scan");
  m.def("l2norm_sq", &l2norm_sq_hip, "squared L2 norm into a device scalar");
  m.def("rmsprop_step", &rmsprop_step_hip, "fused clip+RMSprop on flat buffers");
  m.def("adam_step", &adam_step_hip, "fused clip+Adam on flat buffers");
  m.def("soft_update", &soft_update_hip, "multi-tensor Polyak update");
}
