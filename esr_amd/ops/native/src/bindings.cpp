// Python bindings for the esr_amd gfx950 HIP extension.
#include <torch/extension.h>

// deform_conv.hip
at::Tensor deform_conv2d_forward(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const c10::optional<at::Tensor>&,
    int64_t, int64_t, int64_t, int64_t, int64_t, int64_t, int64_t);
std::vector<at::Tensor> deform_conv2d_backward(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&,
    int64_t, int64_t, int64_t, int64_t, int64_t, int64_t, int64_t);
at::Tensor deform_conv2d_forward_fused(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const c10::optional<at::Tensor>&, int64_t);
at::Tensor deform_im2col_debug(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    int64_t, int64_t, int64_t, int64_t, int64_t, int64_t, int64_t, int64_t,
    int64_t);

// gru_gates.hip
std::vector<at::Tensor> gru_gates_ur_forward(const at::Tensor&,
                                             const at::Tensor&);
std::vector<at::Tensor> gru_gates_ur_backward(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&, const at::Tensor&, const at::Tensor&);
std::vector<at::Tensor> gru_gates_out_forward(const at::Tensor&,
                                              const at::Tensor&,
                                              const at::Tensor&);
std::vector<at::Tensor> gru_gates_out_backward(
    const at::Tensor&, const at::Tensor&, const at::Tensor&,
    const at::Tensor&);

// event_ops.hip
at::Tensor splat_count(const at::Tensor&, int64_t, int64_t);
at::Tensor splat_stack(const at::Tensor&, int64_t, int64_t, int64_t, double,
                       double);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("deform_conv2d_forward", &deform_conv2d_forward,
        "modulated deformable conv forward (gfx950)");
  m.def("deform_conv2d_backward", &deform_conv2d_backward,
        "modulated deformable conv backward (gfx950, batched)");
  m.def("deform_im2col", &deform_im2col_debug,
        "deformable im2col (test hook)");
  m.def("deform_conv2d_forward_fused", &deform_conv2d_forward_fused,
        "fused im2col+MFMA deformable conv forward (test hook)");
  m.def("gru_gates_ur_forward", &gru_gates_ur_forward);
  m.def("gru_gates_ur_backward", &gru_gates_ur_backward);
  m.def("gru_gates_out_forward", &gru_gates_out_forward);
  m.def("gru_gates_out_backward", &gru_gates_out_backward);
  m.def("splat_count", &splat_count);
  m.def("splat_stack", &splat_stack);
  m.attr("gfx_arch") = "gfx950";
}
