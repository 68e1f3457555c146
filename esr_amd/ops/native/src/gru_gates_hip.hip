#include "hip/hip_runtime.h"
// Fused ConvGRU gate kernels for gfx950.
//
// The reference computes the GRU gate math as a chain of separate
// elementwise ops (ESR:models/submodules.py:496-510): 2x sigmoid, tanh,
// 3 muls, 1 sub, 1 add — each a full HBM round trip.  These kernels fuse
// the chain into two passes (one before the out-gate conv, one after),
// each reading its inputs once and writing once.  fp32 and bf16; bf16 is
// loaded vectorized and computed in fp32 (guide G13).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "esr_common.h"

namespace {

// ---- pass 1: (ur_pre [B,2C,H,W], h) -> u = sig(ur[0:C]), r = sig(ur[C:2C]),
//              hr = h*r
template <typename T>
__global__ void gru_ur_fwd_kernel(long long n, long long chw,
                                  const T* __restrict__ ur_pre,
                                  const T* __restrict__ h,
                                  T* __restrict__ u, T* __restrict__ r,
                                  T* __restrict__ hr) {
  ESR_KERNEL_LOOP(i, n) {
    const long long b = i / chw;
    const long long off = i % chw;
    const long long base = b * 2 * chw;
    float uv = esr_sigmoid(esr_to_f32(ur_pre[base + off]));
    float rv = esr_sigmoid(esr_to_f32(ur_pre[base + chw + off]));
    float hv = esr_to_f32(h[i]);
    u[i] = esr_from_f32<T>(uv);
    r[i] = esr_from_f32<T>(rv);
    hr[i] = esr_from_f32<T>(hv * rv);
  }
}

template <typename T>
__global__ void gru_ur_bwd_kernel(long long n, long long chw,
                                  const T* __restrict__ du,
                                  const T* __restrict__ dr,
                                  const T* __restrict__ dhr,
                                  const T* __restrict__ u,
                                  const T* __restrict__ r,
                                  const T* __restrict__ h,
                                  T* __restrict__ d_ur, T* __restrict__ dh) {
  ESR_KERNEL_LOOP(i, n) {
    const long long b = i / chw;
    const long long off = i % chw;
    const long long base = b * 2 * chw;
    float uv = esr_to_f32(u[i]);
    float rv = esr_to_f32(r[i]);
    float hv = esr_to_f32(h[i]);
    float duv = esr_to_f32(du[i]);
    float drv = esr_to_f32(dr[i]);
    float dhrv = esr_to_f32(dhr[i]);
    // d ur_pre[0:C] = du * u(1-u);  d ur_pre[C:2C] = (dr + dhr*h) * r(1-r)
    d_ur[base + off] = esr_from_f32<T>(duv * uv * (1.f - uv));
    d_ur[base + chw + off] =
        esr_from_f32<T>((drv + dhrv * hv) * rv * (1.f - rv));
    dh[i] = esr_from_f32<T>(dhrv * rv);
  }
}

// ---- pass 2: h_new = h*(1-u) + tanh(o_pre)*u ----
template <typename T>
__global__ void gru_out_fwd_kernel(long long n, const T* __restrict__ o_pre,
                                   const T* __restrict__ u,
                                   const T* __restrict__ h,
                                   T* __restrict__ h_new,
                                   T* __restrict__ tanh_o) {
  ESR_KERNEL_LOOP(i, n) {
    float ov = tanhf(esr_to_f32(o_pre[i]));
    float uv = esr_to_f32(u[i]);
    float hv = esr_to_f32(h[i]);
    tanh_o[i] = esr_from_f32<T>(ov);
    h_new[i] = esr_from_f32<T>(hv + uv * (ov - hv));
  }
}

template <typename T>
__global__ void gru_out_bwd_kernel(long long n, const T* __restrict__ dh_new,
                                   const T* __restrict__ u,
                                   const T* __restrict__ h,
                                   const T* __restrict__ tanh_o,
                                   T* __restrict__ do_pre,
                                   T* __restrict__ du, T* __restrict__ dh) {
  ESR_KERNEL_LOOP(i, n) {
    float g = esr_to_f32(dh_new[i]);
    float uv = esr_to_f32(u[i]);
    float hv = esr_to_f32(h[i]);
    float ov = esr_to_f32(tanh_o[i]);
    do_pre[i] = esr_from_f32<T>(g * uv * (1.f - ov * ov));
    du[i] = esr_from_f32<T>(g * (ov - hv));
    dh[i] = esr_from_f32<T>(g * (1.f - uv));
  }
}

#define DISPATCH_ESR_FLOAT(TYPE, NAME, ...)                               \
  [&] {                                                                   \
    if (TYPE == at::kFloat) {                                             \
      using scalar_t = float;                                             \
      return __VA_ARGS__();                                               \
    } else if (TYPE == at::kBFloat16) {                                   \
      using scalar_t = __hip_bfloat16;                                    \
      return __VA_ARGS__();                                               \
    } else {                                                              \
      TORCH_CHECK(false, NAME ": unsupported dtype");                     \
    }                                                                     \
  }()

template <typename T>
T* tp(at::Tensor& t) { return reinterpret_cast<T*>(t.data_ptr()); }
template <typename T>
const T* tcp(const at::Tensor& t) {
  return reinterpret_cast<const T*>(t.data_ptr());
}

}  // namespace

std::vector<at::Tensor> gru_gates_ur_forward(const at::Tensor& ur_pre,
                                             const at::Tensor& h) {
  TORCH_CHECK(ur_pre.is_cuda() && ur_pre.is_contiguous() && h.is_contiguous());
  TORCH_CHECK(ur_pre.size(1) == 2 * h.size(1), "ur_pre must have 2C channels");
  auto u = at::empty_like(h);
  auto r = at::empty_like(h);
  auto hr = at::empty_like(h);
  long long n = h.numel();
  long long chw = n / h.size(0);
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_ESR_FLOAT(h.scalar_type(), "gru_ur_fwd", [&] {
    hipLaunchKernelGGL(gru_ur_fwd_kernel<scalar_t>, dim3(esr_grid(n)),
                       dim3(ESR_BLOCK), 0, stream, n, chw,
                       tcp<scalar_t>(ur_pre), tcp<scalar_t>(h),
                       tp<scalar_t>(u), tp<scalar_t>(r), tp<scalar_t>(hr));
  });
  return {u, r, hr};
}

std::vector<at::Tensor> gru_gates_ur_backward(
    const at::Tensor& du, const at::Tensor& dr, const at::Tensor& dhr,
    const at::Tensor& u, const at::Tensor& r, const at::Tensor& h) {
  auto d_ur = at::empty({h.size(0), 2 * h.size(1), h.size(2), h.size(3)},
                        h.options());
  auto dh = at::empty_like(h);
  long long n = h.numel();
  long long chw = n / h.size(0);
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_ESR_FLOAT(h.scalar_type(), "gru_ur_bwd", [&] {
    hipLaunchKernelGGL(gru_ur_bwd_kernel<scalar_t>, dim3(esr_grid(n)),
                       dim3(ESR_BLOCK), 0, stream, n, chw,
                       tcp<scalar_t>(du), tcp<scalar_t>(dr),
                       tcp<scalar_t>(dhr), tcp<scalar_t>(u), tcp<scalar_t>(r),
                       tcp<scalar_t>(h), tp<scalar_t>(d_ur),
                       tp<scalar_t>(dh));
  });
  return {d_ur, dh};
}

std::vector<at::Tensor> gru_gates_out_forward(const at::Tensor& o_pre,
                                              const at::Tensor& u,
                                              const at::Tensor& h) {
  TORCH_CHECK(o_pre.is_cuda() && o_pre.is_contiguous());
  auto h_new = at::empty_like(h);
  auto tanh_o = at::empty_like(h);
  long long n = h.numel();
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_ESR_FLOAT(h.scalar_type(), "gru_out_fwd", [&] {
    hipLaunchKernelGGL(gru_out_fwd_kernel<scalar_t>, dim3(esr_grid(n)),
                       dim3(ESR_BLOCK), 0, stream, n, tcp<scalar_t>(o_pre),
                       tcp<scalar_t>(u), tcp<scalar_t>(h),
                       tp<scalar_t>(h_new), tp<scalar_t>(tanh_o));
  });
  return {h_new, tanh_o};
}

std::vector<at::Tensor> gru_gates_out_backward(
    const at::Tensor& dh_new, const at::Tensor& u, const at::Tensor& h,
    const at::Tensor& tanh_o) {
  auto do_pre = at::empty_like(h);
  auto du = at::empty_like(h);
  auto dh = at::empty_like(h);
  long long n = h.numel();
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_ESR_FLOAT(h.scalar_type(), "gru_out_bwd", [&] {
    hipLaunchKernelGGL(gru_out_bwd_kernel<scalar_t>, dim3(esr_grid(n)),
                       dim3(ESR_BLOCK), 0, stream, n, tcp<scalar_t>(dh_new),
                       tcp<scalar_t>(u), tcp<scalar_t>(h),
                       tcp<scalar_t>(tanh_o), tp<scalar_t>(do_pre),
                       tp<scalar_t>(du), tp<scalar_t>(dh));
  });
  return {do_pre, du, dh};
}
