#include "hip/hip_runtime.h"
// Fused modulated-deformable-conv FORWARD for gfx950 (CDNA4).
//
// The generic path (deform_conv.hip) materializes the im2col column buffer
// in HBM ([B, C*9, Ho*Wo] fp32) and runs a library GEMM over it.  This
// kernel fuses the two: each workgroup computes a [Cout x 64-pixel] output
// tile for one image, staging 16-deep im2col K-slices in LDS on the fly
// and consuming them with fp32 MFMA (v_mfma_f32_16x16x4_f32 — exact fp32
// at the 157 TF rate, guide §3) — the column buffer never exists.
//
// Scope (the ESRNet alignment shape, ESR:models/model.py:173): kernel 3x3,
// stride 1, pad 1, dilation 1, fp32, Cout % 16 == 0, (C*9) % 16 == 0
// handled by zero-padding the K loop tail; any deformable_groups dividing
// C.  Other configurations dispatch to the generic path.
//
// Geometry: 256 threads = 4 waves.  Wave w owns output rows
// [16w, 16w+16); the block covers min(Cout,64) rows x 64 pixels.  Grid:
// (ceil(HoWo/64), ceil(Cout/64), B).  K loop steps BK=16:
//   stage  B-tile [16][64+1]   im2col values (4 per thread, on the fly)
//   stage  A-tile [64][16+1]   weight slice
//   4x     mfma_f32_16x16x4_f32 per (wave, n-subtile) accumulating 16x16.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "esr_common.h"

namespace {

using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int BM = 64;   // output-channel tile
constexpr int BN = 64;   // pixel tile
constexpr int BK = 16;   // K slice
constexpr int KTAPS = 9; // 3x3

__global__ __launch_bounds__(256)
void dcn_fused_fwd_kernel(
    const float* __restrict__ im, const float* __restrict__ offset,
    const float* __restrict__ mask, const float* __restrict__ weight,
    const float* __restrict__ bias, float* __restrict__ out,
    int B, int C, int H, int W, int Cout, int dg) {
  const int HoWo = H * W;            // stride1/pad1: out dims == in dims
  const int K = C * KTAPS;
  const int cpg = C / dg;

  const int pix0 = blockIdx.x * BN;
  const int m0 = blockIdx.y * BM;
  const int b = blockIdx.z;

  // double-buffered LDS: sample slice k0+16 while MFMA consumes slice k0
  __shared__ float Bt[2][BK][BN + 1];   // im2col slices, +1 col pad
  __shared__ float At[2][BM][BK + 1];   // weight slices

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  // accumulators: wave covers rows [16*wave, 16*wave+16) x 4 pixel subtiles
  f32x4 acc[4] = {{0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}, {0, 0, 0, 0}};

  const float* im_b = im + (long long)b * C * H * W;
  const float* off_b = offset + (long long)b * dg * 2 * KTAPS * HoWo;
  const float* msk_b = mask + (long long)b * dg * KTAPS * HoWo;

  auto stage = [&](int k0, int buf) {
    // ---- stage B-tile: each thread samples 4 im2col values ----
    #pragma unroll
    for (int kr = wave; kr < BK; kr += 4) {
      const int k = k0 + kr;
      float val = 0.f;
      const int px = pix0 + lane;
      if (k < K && px < HoWo) {
        const int c = k / KTAPS;
        const int tap = k - c * KTAPS;
        const int ti = tap / 3, tj = tap - ti * 3;
        const int grp = c / cpg;
        const int ho = px / W, wo = px - ho * W;
        const long long obase = ((long long)grp * 2 * KTAPS) * HoWo + px;
        const float off_h = off_b[obase + (2 * tap) * HoWo];
        const float off_w = off_b[obase + (2 * tap + 1) * HoWo];
        const float m = msk_b[((long long)grp * KTAPS + tap) * HoWo + px];
        const float h_im = ho - 1 + ti + off_h;
        const float w_im = wo - 1 + tj + off_w;
        if (h_im > -1.f && w_im > -1.f && h_im < H && w_im < W) {
          const float* imc = im_b + (long long)c * H * W;
          const int h0 = (int)floorf(h_im);
          const int w0 = (int)floorf(w_im);
          const float lh = h_im - h0, lw = w_im - w0;
          float v00 = (h0 >= 0 && w0 >= 0) ? imc[h0 * W + w0] : 0.f;
          float v01 = (h0 >= 0 && w0 + 1 < W) ? imc[h0 * W + w0 + 1] : 0.f;
          float v10 = (h0 + 1 < H && w0 >= 0) ? imc[(h0 + 1) * W + w0] : 0.f;
          float v11 = (h0 + 1 < H && w0 + 1 < W) ? imc[(h0 + 1) * W + w0 + 1] : 0.f;
          val = ((1 - lh) * (1 - lw) * v00 + (1 - lh) * lw * v01 +
                 lh * (1 - lw) * v10 + lh * lw * v11) * m;
        }
      }
      Bt[buf][kr][lane] = val;
    }
    // ---- stage A-tile: weight[m0+row][k0+kc] (4 values per thread) ----
    #pragma unroll
    for (int i = tid; i < BM * BK; i += 256) {
      const int row = i >> 4;         // /BK
      const int kc = i & 15;
      const int gm = m0 + row;
      const int gk = k0 + kc;
      At[buf][row][kc] = (gm < Cout && gk < K)
          ? weight[(long long)gm * K + gk] : 0.f;
    }
  };

  stage(0, 0);
  __syncthreads();
  int cur = 0;
  for (int k0 = 0; k0 < K; k0 += BK) {
    if (k0 + BK < K)
      stage(k0 + BK, cur ^ 1);        // overlaps the MFMAs below
    // ---- MFMA: 4 k-steps of 4, 4 pixel subtiles ----
    #pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      const int kk = ks * 4 + (lane >> 4);          // this lane's k
      const float a = At[cur][wave * 16 + (lane & 15)][kk];
      #pragma unroll
      for (int nt = 0; nt < 4; ++nt) {
        const float bv = Bt[cur][kk][nt * 16 + (lane & 15)];
        acc[nt] = __builtin_amdgcn_mfma_f32_16x16x4f32(a, bv, acc[nt], 0, 0, 0);
      }
    }
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: D[i=row][j=col], col = lane&15, row = (lane>>4)*4 + r ----
  const int col_l = lane & 15;
  const int row_base = (lane >> 4) * 4;
  #pragma unroll
  for (int nt = 0; nt < 4; ++nt) {
    const int px = pix0 + nt * 16 + col_l;
    if (px >= HoWo) continue;
    #pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = m0 + wave * 16 + row_base + r;
      if (m < Cout) {
        float v = acc[nt][r];
        if (bias != nullptr) v += bias[m];
        out[((long long)b * Cout + m) * HoWo + px] = v;
      }
    }
  }
}

}  // namespace

bool dcn_fused_applicable(const at::Tensor& input, const at::Tensor& weight,
                          int64_t sh, int64_t sw, int64_t ph, int64_t pw,
                          int64_t dh, int64_t dw, int64_t dg) {
  return input.scalar_type() == at::kFloat && weight.size(2) == 3 &&
         weight.size(3) == 3 && sh == 1 && sw == 1 && ph == 1 && pw == 1 &&
         dh == 1 && dw == 1 && weight.size(0) % 16 == 0 &&
         input.size(1) % dg == 0;
}

at::Tensor deform_conv2d_forward_fused(
    const at::Tensor& input, const at::Tensor& offset, const at::Tensor& mask,
    const at::Tensor& weight, const c10::optional<at::Tensor>& bias,
    int64_t dg) {
  const int B = input.size(0), C = input.size(1);
  const int H = input.size(2), W = input.size(3);
  const int Cout = weight.size(0);
  auto out = at::empty({B, Cout, H, W}, input.options());
  dim3 grid((H * W + BN - 1) / BN, (Cout + BM - 1) / BM, B);
  auto stream = at::hip::getCurrentHIPStream();
  const float* bias_ptr = (bias.has_value() && bias->defined())
      ? bias->data_ptr<float>() : nullptr;
  hipLaunchKernelGGL(dcn_fused_fwd_kernel, grid, dim3(256), 0, stream,
                     input.data_ptr<float>(), offset.data_ptr<float>(),
                     mask.data_ptr<float>(), weight.data_ptr<float>(),
                     bias_ptr, out.data_ptr<float>(),
                     B, C, H, W, Cout, (int)dg);
  return out;
}
