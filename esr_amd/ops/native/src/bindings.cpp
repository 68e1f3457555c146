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

// conv2d.hip
at::Tensor conv2d_fwd_mfma(const at::Tensor&, const at::Tensor&,
                           const c10::optional<at::Tensor>&,
                           int64_t, int64_t, int64_t, int64_t);
at::Tensor conv2d_fwd_valu(const at::Tensor&, const at::Tensor&,
                           const c10::optional<at::Tensor>&, int64_t, int64_t);
at::Tensor conv2d_fwd_valu2(const at::Tensor&, const at::Tensor&,
                            const c10::optional<at::Tensor>&, int64_t, int64_t);
at::Tensor conv2d_dgrad_s2(const at::Tensor&, const at::Tensor&,
                           int64_t, int64_t);
at::Tensor conv2d_wgrad_mfma(const at::Tensor&, const at::Tensor&,
                             int64_t, int64_t, int64_t, int64_t);
at::Tensor act_grad(const at::Tensor&, const at::Tensor&, int64_t);
at::Tensor gemm16_probe(const at::Tensor&, const at::Tensor&);

// redistribute.hip
std::vector<at::Tensor> redistribute_stack_hip(const at::Tensor&, int64_t,
                                               int64_t, int64_t, int64_t);

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
  m.def("conv2d_fwd_mfma", &conv2d_fwd_mfma,
        "bf16 NCHW conv fwd, MFMA implicit GEMM, fused bias+act (gfx950)");
  m.def("conv2d_fwd_valu", &conv2d_fwd_valu,
        "bf16 NCHW conv fwd, direct VALU, fused bias+act (gfx950)");
  m.def("conv2d_fwd_valu2", &conv2d_fwd_valu2,
        "bf16 NCHW conv fwd, register-strip VALU v2 for tiny channels");
  m.def("conv2d_dgrad_s2", &conv2d_dgrad_s2,
        "bf16 stride-2 conv input-grad (gfx950)");
  m.def("conv2d_wgrad_mfma", &conv2d_wgrad_mfma,
        "bf16 conv weight-grad, MFMA split-K + fp32 atomics (gfx950)");
  m.def("act_grad", &act_grad, "fused activation backward (bf16)");
  m.def("gemm16_probe", &gemm16_probe, "16x16x32 bf16 MFMA fragment probe");
  m.def("redistribute_stack_hip", &redistribute_stack_hip,
        "count stack -> sorted event cloud, device-only "
        "(scan + scatter + segmented radix sort)");
  m.attr("gfx_arch") = "gfx950";
}
