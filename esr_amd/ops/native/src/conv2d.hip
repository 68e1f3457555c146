// Hand-written gfx950 (CDNA4) direct convolution kernels, bf16 NCHW.
//
// Replaces the MIOpen conv + NCHW<->NHWC batched_transpose + autocast cast
// chain for the shapes ESRNet actually runs (3x3 s1/s2 p1 and 1x1 convs at
// C in {2..192}, reference ConvLayer ESR:models/submodules.py:159-200 and
// ConvGRU convs :474-514): bf16-native in/out, fp32 accumulate, fused
// bias + ReLU/sigmoid/tanh epilogues, no layout round trips.
//
// Two forward paths, picked by the host by shape:
//   * MFMA implicit GEMM (mfma_f32_16x16x32_bf16): M = Cout tile (32/64),
//     N = 64 output pixels (2 rows x 32 cols), K = Cin in chunks of 32,
//     3x3 taps looped outside K.  The input patch is staged in LDS in a
//     pixel-major / channel-contiguous layout (40-short slots = 80 B so
//     ds_read_b128 lane groups hit distinct banks; guide §2/G4), weights
//     are read as packed [tap][Cout_p][Cin_p] fragments straight from L2.
//   * VALU direct kernel for bandwidth-bound small-channel shapes
//     (Cin small or Cout < 16 where MFMA tiles would idle).
//
// Backward: stride-1 input-grad IS this forward with flipped/transposed
// packed weights (host side); stride-2 input-grad has a dedicated
// scatter-form VALU kernel; stride-1 weight-grad is the MFMA kernel with
// a DETERMINISTIC two-stage reduction (conv2d_wgrad_s1_kernel below),
// stride-2 weight-grad the older per-row-tile variant with transposed
// line-parallel atomics.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>
#include "esr_common.h"

namespace {

using s16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

constexpr int TW = 32;    // output-pixel tile cols
constexpr int TH = 2;     // output-pixel tile rows
constexpr int CIK = 32;   // input channels per K chunk
constexpr int SLOT = 40;  // shorts per pixel slot (32 ci + 8 pad = 80 B:
                          // 20-dword stride, gcd(20,64)=4 -> the 16-lane
                          // ds_read_b128 groups see 16 distinct banks)

template <int ACT> ESR_INLINE float apply_act(float v);
template <> ESR_INLINE float apply_act<0>(float v) { return v; }
template <> ESR_INLINE float apply_act<1>(float v) { return v > 0.f ? v : 0.f; }
template <> ESR_INLINE float apply_act<2>(float v) { return esr_sigmoid(v); }
template <> ESR_INLINE float apply_act<3>(float v) { return tanhf(v); }

ESR_INLINE float bf16u_to_f(ushort u) {
  unsigned int x = (unsigned int)u << 16;
  return __uint_as_float(x);
}
ESR_INLINE ushort f_to_bf16u(float f) {
  // round-to-nearest-even, matching torch's float->bf16 cast
  unsigned int x = __float_as_uint(f);
  unsigned int lsb = (x >> 16) & 1;
  x += 0x7fffu + lsb;
  return (ushort)(x >> 16);
}

// ---------------------------------------------------------------------------
// MFMA forward: one block = [32*MREP cout] x [64 pixels] for one frame.
// Wave grid 2(M) x 2(N); per wave MREP m-frags x 2 n-frags of 16x16.
// ---------------------------------------------------------------------------

template <int KS, int STRIDE, int ACT, int MREP>
__global__ __launch_bounds__(256)
void conv2d_fwd_mfma_kernel(
    const ushort* __restrict__ x,      // [B, Cin, H, W] bf16
    const ushort* __restrict__ wp,     // [KS*KS, Cout_p, Cin_p] bf16 packed
    const float* __restrict__ bias,    // [Cout] fp32 or nullptr
    ushort* __restrict__ y,            // [B, Cout, Ho, Wo] bf16
    int Cin, int H, int W, int Cout, int Ho, int Wo,
    int Cin_p, int Cout_p, int ntx) {
  constexpr int PH = STRIDE * TH + (KS - 1);
  constexpr int PW = STRIDE * TW + (KS - 1);
  constexpr int PAD = KS / 2;
  __shared__ ushort patch[PH * PW * SLOT];

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wm = wave >> 1, wn = wave & 1;

  const int ty0 = (blockIdx.x / ntx) * TH;
  const int tx0 = (blockIdx.x % ntx) * TW;
  const int b = blockIdx.y;
  const int co0 = blockIdx.z * (32 * MREP);

  const long long x_b = (long long)b * Cin * H * W;
  const long long plane = (long long)H * W;
  const int in_y0 = ty0 * STRIDE - PAD;
  const int in_x0 = tx0 * STRIDE - PAD;

  f32x4 acc[MREP][2] = {};

  const int kgrp = (lane >> 4) * 8;  // k-offset of this lane's fragment rows
  const int nchunks = Cin_p / CIK;
  for (int ck = 0; ck < nchunks; ++ck) {
    const int ci0 = ck * CIK;
    if (ck) __syncthreads();
    // ---- stage: thread <-> patch pixel; 32 plane-strided reads per pixel
    // are wave-coalesced (consecutive threads read consecutive ix of the
    // same channel plane); 4x ds_write_b128 per pixel, conflict-free slots.
    for (int p = tid; p < PH * PW; p += 256) {
      const int py = p / PW, px = p % PW;
      const int iy = in_y0 + py, ix = in_x0 + px;
      ushort vals[CIK];
      if (iy >= 0 && iy < H && ix >= 0 && ix < W) {
        const ushort* src = x + x_b + ci0 * plane + (long long)iy * W + ix;
        const int cmax = Cin - ci0;  // >= 1 here (ci0 < Cin or full pad)
#pragma unroll
        for (int c = 0; c < CIK; ++c)
          vals[c] = (c < cmax) ? src[c * plane] : (ushort)0;
      } else {
#pragma unroll
        for (int c = 0; c < CIK; ++c) vals[c] = 0;
      }
#pragma unroll
      for (int v = 0; v < CIK / 8; ++v)
        *reinterpret_cast<s16x8*>(&patch[p * SLOT + v * 8]) =
            *reinterpret_cast<const s16x8*>(&vals[v * 8]);
    }
    __syncthreads();

    // ---- MFMA: 9 (or 1) taps x MREP x 2 fragments per wave per chunk
#pragma unroll
    for (int tap = 0; tap < KS * KS; ++tap) {
      const int ky = tap / KS, kx = tap % KS;
      s16x8 a[MREP];
#pragma unroll
      for (int m = 0; m < MREP; ++m) {
        const int row = co0 + wm * 16 * MREP + m * 16 + (lane & 15);
        a[m] = *reinterpret_cast<const s16x8*>(
            &wp[((long long)tap * Cout_p + row) * Cin_p + ci0 + kgrp]);
      }
#pragma unroll
      for (int nf = 0; nf < 2; ++nf) {
        const int pidx = wn * 32 + nf * 16 + (lane & 15);
        const int py = (pidx >> 5) * STRIDE + ky;
        const int px = (pidx & 31) * STRIDE + kx;
        const s16x8 bfrag = *reinterpret_cast<const s16x8*>(
            &patch[(py * PW + px) * SLOT + kgrp]);
#pragma unroll
        for (int m = 0; m < MREP; ++m)
          acc[m][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a[m], bfrag, acc[m][nf], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: bias + activation fused, bf16 stores
  // D layout (guide §3): col = lane&15 (pixel), row = (lane>>4)*4 + j (cout)
#pragma unroll
  for (int m = 0; m < MREP; ++m) {
#pragma unroll
    for (int nf = 0; nf < 2; ++nf) {
      const int pidx = wn * 32 + nf * 16 + (lane & 15);
      const int oy = ty0 + (pidx >> 5), ox = tx0 + (pidx & 31);
      if (oy >= Ho || ox >= Wo) continue;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int co = co0 + wm * 16 * MREP + m * 16 + (lane >> 4) * 4 + j;
        if (co >= Cout) continue;
        float v = acc[m][nf][j];
        if (bias) v += bias[co];
        v = apply_act<ACT>(v);
        y[(((long long)b * Cout + co) * Ho + oy) * Wo + ox] = f_to_bf16u(v);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// VALU direct forward: grid-stride over outputs; for bandwidth-bound
// small-channel shapes (head/tail/attention convs).  Neighbouring lanes
// read neighbouring ix -> coalesced; weights broadcast through K$/L1.
// ---------------------------------------------------------------------------

template <int KS, int STRIDE, int ACT>
__global__ void conv2d_fwd_valu_kernel(
    long long n, const ushort* __restrict__ x,
    const ushort* __restrict__ w,      // [Cout, Cin, KS, KS] bf16 (unpacked)
    const float* __restrict__ bias, ushort* __restrict__ y,
    int Cin, int H, int W, int Cout, int Ho, int Wo) {
  constexpr int PAD = KS / 2;
  ESR_KERNEL_LOOP(i, n) {
    const int ox = (int)(i % Wo);
    long long t = i / Wo;
    const int oy = (int)(t % Ho); t /= Ho;
    const int co = (int)(t % Cout);
    const int b = (int)(t / Cout);
    const long long plane = (long long)H * W;
    const ushort* xb = x + (long long)b * Cin * plane;
    const ushort* wc = w + (long long)co * Cin * KS * KS;
    float acc = bias ? bias[co] : 0.f;
    const int iy0 = oy * STRIDE - PAD, ix0 = ox * STRIDE - PAD;
    for (int ci = 0; ci < Cin; ++ci) {
      const ushort* xp = xb + ci * plane;
      const ushort* wk = wc + ci * KS * KS;
#pragma unroll
      for (int ky = 0; ky < KS; ++ky) {
        const int iy = iy0 + ky;
        if (iy < 0 || iy >= H) continue;
#pragma unroll
        for (int kx = 0; kx < KS; ++kx) {
          const int ix = ix0 + kx;
          if (ix < 0 || ix >= W) continue;
          acc += bf16u_to_f(xp[(long long)iy * W + ix]) *
                 bf16u_to_f(wk[ky * KS + kx]);
        }
      }
    }
    y[i] = f_to_bf16u(apply_act<ACT>(acc));
  }
}

// ---------------------------------------------------------------------------
// VALU forward v2 for the tiny-channel shapes (head 2->8, enc1 8->16,
// tail 8->2 at 256x256): each thread computes an 8-wide output strip for
// up to 8 cout at once — input row segments live in registers and are
// reused across cout and taps, weights broadcast from LDS.  The v1
// per-output kernel was instruction-bound (one bounds-checked scalar load
// per MAC); this one does ~18 loads per 1152 MACs on the head shape.
// ---------------------------------------------------------------------------

template <int KS, int STRIDE, int ACT, int COCH>
__global__ __launch_bounds__(256)
void conv2d_fwd_valu2_kernel(
    const ushort* __restrict__ x,   // [B, Cin, H, W] bf16
    const ushort* __restrict__ w,   // [Cout, Cin, KS, KS] bf16
    const float* __restrict__ bias, ushort* __restrict__ y,
    int Cin, int H, int W, int Cout, int Ho, int Wo,
    long long nstrips, int nsx) {
  constexpr int PAD = KS / 2;
  constexpr int VEC = 8;                       // outputs per thread strip
  constexpr int NPX = VEC * STRIDE + KS - 1;   // input px per row segment
  __shared__ ushort wl[COCH * 32 * KS * KS];   // this co-chunk x <=32ci

  // ---- preload this co-chunk's weights once per block
  const int co0 = blockIdx.z * COCH;
  const int nw = COCH * Cin * KS * KS;
  for (int i = threadIdx.x; i < nw; i += 256) {
    const int co = i / (Cin * KS * KS);
    wl[i] = (co0 + co < Cout)
        ? w[((long long)(co0 + co) * Cin) * KS * KS + (i % (Cin * KS * KS))]
        : (ushort)0;
  }
  __syncthreads();

  for (long long s = (long long)blockIdx.x * blockDim.x + threadIdx.x;
       s < nstrips; s += (long long)gridDim.x * blockDim.x) {
    const int sx = (int)(s % nsx);
    const int oy = (int)((s / nsx) % Ho);
    const int b = (int)(s / ((long long)nsx * Ho));
    const int ox0 = sx * VEC;
    const long long plane = (long long)H * W;
    const ushort* xb = x + (long long)b * Cin * plane;

    float acc[COCH][VEC];   // compile-time indexed everywhere (rule 20)
#pragma unroll
    for (int c = 0; c < COCH; ++c)
#pragma unroll
      for (int j = 0; j < VEC; ++j) acc[c][j] = 0.f;

    const int ix0 = ox0 * STRIDE - PAD;
    const bool interior = ix0 >= 0 && ix0 + NPX <= W;
    for (int ci = 0; ci < Cin; ++ci) {
      const ushort* xp = xb + ci * plane;
#pragma unroll
      for (int ky = 0; ky < KS; ++ky) {
        const int iy = oy * STRIDE + ky - PAD;
        float seg[NPX];
        if (iy >= 0 && iy < H) {
          const ushort* row = xp + (long long)iy * W;
          if (interior) {
#pragma unroll
            for (int e = 0; e < NPX; ++e)
              seg[e] = bf16u_to_f(row[ix0 + e]);
          } else {
#pragma unroll
            for (int e = 0; e < NPX; ++e) {
              const int ix = ix0 + e;
              seg[e] = (ix >= 0 && ix < W) ? bf16u_to_f(row[ix]) : 0.f;
            }
          }
        } else {
#pragma unroll
          for (int e = 0; e < NPX; ++e) seg[e] = 0.f;
        }
        const ushort* wrow = &wl[(ci * KS + ky) * KS];
#pragma unroll
        for (int c = 0; c < COCH; ++c) {
          float wk[KS];
#pragma unroll
          for (int kx = 0; kx < KS; ++kx)
            wk[kx] = bf16u_to_f(wrow[c * Cin * KS * KS + kx]);
#pragma unroll
          for (int j = 0; j < VEC; ++j) {
            float v = acc[c][j];
#pragma unroll
            for (int kx = 0; kx < KS; ++kx)
              v = fmaf(seg[j * STRIDE + kx], wk[kx], v);
            acc[c][j] = v;
          }
        }
      }
    }

    // ---- epilogue: bias + act, vector bf16 stores per cout row
    const int nvalid = Wo - ox0 < VEC ? Wo - ox0 : VEC;
#pragma unroll
    for (int c = 0; c < COCH; ++c) {
      const int co = co0 + c;
      if (co >= Cout) break;
      ushort packed[VEC];
#pragma unroll
      for (int j = 0; j < VEC; ++j) {
        float v = acc[c][j];
        if (bias) v += bias[co];
        packed[j] = f_to_bf16u(apply_act<ACT>(v));
      }
      ushort* dst = y + (((long long)b * Cout + co) * Ho + oy) * Wo + ox0;
      // vector store only when the row base is 16B-aligned (Wo % 8 != 0
      // leaves interior strips misaligned)
      if (nvalid == VEC && (((unsigned long long)(uintptr_t)dst) & 15) == 0) {
        *reinterpret_cast<s16x8*>(dst) =
            *reinterpret_cast<const s16x8*>(packed);
      } else {
        for (int j = 0; j < nvalid; ++j) dst[j] = packed[j];
      }
    }
  }
}

// ---------------------------------------------------------------------------
// Stride-2 input-grad (scatter form as a gather): dx[iy][ix] sums the <=4
// taps whose (iy + PAD - ky) is even, over all Cout.
// ---------------------------------------------------------------------------

template <int KS>
__global__ void conv2d_dgrad_s2_valu_kernel(
    long long n, const ushort* __restrict__ dy,  // [B, Cout, Ho, Wo]
    const ushort* __restrict__ w,                // [Cout, Cin, KS, KS]
    ushort* __restrict__ dx,                     // [B, Cin, H, W]
    int Cin, int H, int W, int Cout, int Ho, int Wo) {
  constexpr int PAD = KS / 2;
  ESR_KERNEL_LOOP(i, n) {
    const int ix = (int)(i % W);
    long long t = i / W;
    const int iy = (int)(t % H); t /= H;
    const int ci = (int)(t % Cin);
    const int b = (int)(t / Cin);
    float acc = 0.f;
#pragma unroll
    for (int ky = 0; ky < KS; ++ky) {
      const int ty = iy + PAD - ky;
      if (ty < 0 || (ty & 1)) continue;
      const int oy = ty >> 1;
      if (oy >= Ho) continue;
#pragma unroll
      for (int kx = 0; kx < KS; ++kx) {
        const int tx = ix + PAD - kx;
        if (tx < 0 || (tx & 1)) continue;
        const int ox = tx >> 1;
        if (ox >= Wo) continue;
        for (int co = 0; co < Cout; ++co) {
          acc += bf16u_to_f(dy[(((long long)b * Cout + co) * Ho + oy) * Wo + ox]) *
                 bf16u_to_f(w[(((long long)co * Cin + ci) * KS + ky) * KS + kx]);
        }
      }
    }
    dx[i] = f_to_bf16u(acc);
  }
}

// ---------------------------------------------------------------------------
// MFMA weight-grad v1 (kept for STRIDE 2), split-K over output pixels.
// dW[co][ci][ky][kx] = sum_pix dpre[co][oy][ox] * X[ci][oy*s+ky-1][ox*s+kx-1]
//   A = dpre, pre-shifted per kx (3 small scalar-staged copies, halo'd)
//   B = X, one aligned vector-staged copy [ci][ky-row][u-chunk]
// Each wave owns its own LDS tiles and an independent slice of output
// rows; the flush reduces the 4 waves through LDS then issues atomics in
// the TRANSPOSED [Cin_p][Cout_p] scratch (lanes span cachelines).
// ---------------------------------------------------------------------------

constexpr int WG_PW = 40;  // padded 32-px rows, same bank math as SLOT

template <int KS, int STRIDE>
__global__ __launch_bounds__(256)
void conv2d_wgrad_mfma_kernel(
    const ushort* __restrict__ x,     // [B, Cin, H, W]
    const ushort* __restrict__ dpre,  // [B, Cout, Ho, Wo]
    float* __restrict__ dwp,          // TRANSPOSED [Cin_p, Cout_p, KS, KS]
    int Cin, int H, int W, int Cout, int Ho, int Wo,
    int Cin_p, int Cout_p, int rows_per_blk) {
  constexpr int PAD = KS / 2;
  constexpr int NTAP = KS * KS;
  // per-wave LDS: X [16ci][KS rows][WG_PW] + dpre shifted copies [KS][16co][WG_PW]
  __shared__ ushort xs_all[4][16 * KS * WG_PW];
  __shared__ ushort dp_all[4][KS * 16 * WG_PW];
  __shared__ float red[16 * 16];  // cross-wave reduction buffer (per tap)

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  ushort* xs = xs_all[wave];
  ushort* dp = dp_all[wave];

  const int co0 = blockIdx.z * 16;
  const int ci0 = blockIdx.y * 16;
  const long long nslab = blockIdx.x;
  const int uw = (W + TW - 1) / TW;        // u-chunks per input row
  const int b = (int)(nslab / ((Ho + rows_per_blk - 1) / rows_per_blk * uw));
  const int rslab = (int)(nslab % ((Ho + rows_per_blk - 1) / rows_per_blk * uw));
  const int r0 = (rslab / uw) * rows_per_blk;
  const int ux0 = (rslab % uw) * TW;

  const long long xplane = (long long)H * W;
  const long long dplane = (long long)Ho * Wo;
  const ushort* xb = x + (long long)b * Cin * xplane;
  const ushort* db = dpre + (long long)b * Cout * dplane;

  f32x4 acc[NTAP] = {};
  const int kgrp = (lane >> 4) * 8;

  for (int oy = r0 + wave; oy < min(r0 + rows_per_blk, Ho); oy += 4) {
    // ---- stage X rows iy = oy*s + ky - PAD at u in [ux0, ux0+32), aligned
    // lane <-> (ci, ky, u-chunk-of-8): 16*KS*4 units
    for (int u = lane; u < 16 * KS * 4; u += 64) {
      const int ci = u / (KS * 4);
      const int ky = (u / 4) % KS;
      const int c8 = u % 4;
      const int iy = oy * STRIDE + ky - PAD;
      ushort vals[8];
      const int gci = ci0 + ci;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int ix = ux0 + c8 * 8 + e;
        vals[e] = (gci < Cin && iy >= 0 && iy < H && ix < W)
            ? xb[gci * xplane + (long long)iy * W + ix] : (ushort)0;
      }
      *reinterpret_cast<s16x8*>(&xs[(ci * KS + ky) * WG_PW + c8 * 8]) =
          *reinterpret_cast<const s16x8*>(&vals[0]);
    }
    // ---- stage dpre shifted per kx: dp[kx][co][j] = dpre[co][oy][ox(u)]
    // u = ux0 + j; ox = (u + PAD - kx) / STRIDE when divisible, else 0
    for (int u = lane; u < KS * 16 * 4; u += 64) {
      const int kx = u / (16 * 4);
      const int co = (u / 4) % 16;
      const int c8 = u % 4;
      const int gco = co0 + co;
      ushort vals[8];
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int uu = ux0 + c8 * 8 + e;
        const int tx = uu + PAD - kx;
        ushort v = 0;
        if (gco < Cout && tx >= 0 && (STRIDE == 1 || (tx & 1) == 0)) {
          const int ox = tx / STRIDE;
          if (ox < Wo)
            v = db[gco * dplane + (long long)oy * Wo + ox];
        }
        vals[e] = v;
      }
      *reinterpret_cast<s16x8*>(&dp[(kx * 16 + co) * WG_PW + c8 * 8]) =
          *reinterpret_cast<const s16x8*>(&vals[0]);
    }
    // wave-synchronous: the same wave wrote and now reads its own LDS
    // slices, but the writer/reader LANES differ, so the compiler's
    // per-thread scoreboard can't see the dependency — drain the LDS queue
    // explicitly (no barrier needed: wave64 executes in lockstep)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- 2 K-subchunks of 16 pixels... K = 32 pixels of this row chunk
#pragma unroll
    for (int ky = 0; ky < KS; ++ky) {
#pragma unroll
      for (int kx = 0; kx < KS; ++kx) {
        const s16x8 a = *reinterpret_cast<const s16x8*>(
            &dp[(kx * 16 + (lane & 15)) * WG_PW + kgrp]);
        const s16x8 bf = *reinterpret_cast<const s16x8*>(
            &xs[((lane & 15) * KS + ky) * WG_PW + kgrp]);
        acc[ky * KS + kx] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a, bf, acc[ky * KS + kx], 0, 0, 0);
      }
    }
  }

  // ---- flush: D col = ci = lane&15, row = co = (lane>>4)*4 + j.
  // 4-wave partials meet in LDS (4-way atomic contention max), then one
  // global fp32 atomic per dW element per block (guide G12).
  for (int tap = 0; tap < NTAP; ++tap) {
    const int ky = tap / KS, kx = tap % KS;
    red[tid] = 0.f;
    __syncthreads();
#pragma unroll
    for (int j = 0; j < 4; ++j)
      atomicAdd(&red[((lane >> 4) * 4 + j) * 16 + (lane & 15)], acc[tap][j]);
    __syncthreads();
    if (tid < 256) {
      // consecutive threads vary ci -> distinct lines in the transposed
      // [Cin_p][Cout_p] scratch
      const int co = tid >> 4, ci = tid & 15;
      atomicAdd(&dwp[(((long long)(ci0 + ci) * Cout_p + co0 + co) * KS + ky)
                     * KS + kx], red[tid]);
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// MFMA weight-grad v2, stride-1 (3x3 p1 and 1x1): block-wide staged tiles,
// all staging as aligned b128 vectors, kx-shifts resolved by REGISTER
// shuffles of adjacent aligned fragments (no shifted LDS copies, no
// unaligned ds_read — guide G17).
//
//   block tile: [64 co] x [32 ci] x taps;  wave w owns co slice w*16.
//   A = dpre rows (halo'd u-domain), B = X rows; D[row=co][col=ci].
//   K = output pixels, chunked 32 per output row; fp32 atomics out.
// ---------------------------------------------------------------------------

constexpr int WGV2_DPW = 56;  // dp row stride in shorts (112 B: 28-dword
                              // stride, gcd(28,64)=4 -> conflict-free b128)
constexpr int WGV2_XW = 40;   // x row stride (same bank math as SLOT)

template <int KS>
__global__ __launch_bounds__(256)
void conv2d_wgrad_s1_kernel(
    const ushort* __restrict__ x,     // [B, Cin, H, W]
    const ushort* __restrict__ dpre,  // [B, Cout, H, W] (stride 1: Ho=H)
    float* __restrict__ dwp,          // partial tiles [g][32ci][64co][NTAP]
    int Cin, int H, int W, int Cout,
    int Cin_p, int Cout_p, int uw, int B, int fpb, int abl) {
  // abl (ablation, tools/bench_conv.py --wgrad): 0 = full, 1 = skip the
  // flush, 2 = skip MFMA (staging only; flush still runs)
  constexpr int PAD = KS / 2;
  constexpr int NTAP = KS * KS;
  constexpr int RB = 4;                       // output rows per barrier pair
  __shared__ ushort dp[RB * 64 * WGV2_DPW];   // px domain [-8, 48) per row
  __shared__ ushort xs[32 * (RB + 2) * WGV2_XW];  // rolling 6-row window

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;

  const int b0 = (blockIdx.x / uw) * fpb;
  const int ux0 = (blockIdx.x % uw) * TW;
  const int ci0 = blockIdx.y * 32;
  const int co0 = blockIdx.z * 64;

  const long long plane = (long long)H * W;

  f32x4 acc[NTAP][2] = {};
  const int kgrp = (lane >> 4) * 8;
  constexpr int NSLOT = RB + 2;

  // stage one X row (iy) into rolling slot (iy mod NSLOT); zeros when oob
  auto stage_x_row = [&](const ushort* xb, int iy) {
    const int slot = ((iy % NSLOT) + NSLOT) % NSLOT;
    for (int u = tid; u < 32 * 4; u += 256) {
      const int ci = u >> 2, blk = u & 3;
      const int gci = ci0 + ci;
      const int px0 = ux0 + blk * 8;
      ushort vals[8] = {};
      if (gci < Cin && iy >= 0 && iy < H) {
        const ushort* src = xb + gci * plane + (long long)iy * W;
        if (px0 + 8 <= W) {
          *reinterpret_cast<s16x8*>(vals) =
              *reinterpret_cast<const s16x8*>(src + px0);
        } else {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            vals[e] = (px0 + e < W) ? src[px0 + e] : (ushort)0;
        }
      }
      *reinterpret_cast<s16x8*>(&xs[(ci * NSLOT + slot) * WGV2_XW + blk * 8]) =
          *reinterpret_cast<const s16x8*>(vals);
    }
  };

  for (int bf = 0; bf < fpb && b0 + bf < B; ++bf) {
    const int b = b0 + bf;
    const ushort* xb = x + (long long)b * Cin * plane;
    const ushort* db = dpre + (long long)b * Cout * plane;

    // prologue: window rows -1, 0 (zeros via oob for -1)
    __syncthreads();
    if (KS == 3) {
      stage_x_row(xb, -1);
      stage_x_row(xb, 0);
    }

    for (int oy0 = 0; oy0 < H; oy0 += RB) {
      // ---- stage RB dp rows [64 co] at px [-8, 48): 7 b128 per row
      // (the kx=0 fragment at kgrp=24 reaches one pixel past the chunk)
      for (int u = tid; u < RB * 64 * 7; u += 256) {
        const int r = u / (64 * 7);
        const int co = (u / 7) % 64, blk = u % 7;
        const int gco = co0 + co;
        const int oy = oy0 + r;
        const int px0 = ux0 - 8 + blk * 8;
        ushort vals[8] = {};
        if (gco < Cout && oy < H) {
          const ushort* src = db + gco * plane + (long long)oy * W;
          if (px0 >= 0 && px0 + 8 <= W) {
            *reinterpret_cast<s16x8*>(vals) =
                *reinterpret_cast<const s16x8*>(src + px0);
          } else {
#pragma unroll
            for (int e = 0; e < 8; ++e) {
              const int px = px0 + e;
              vals[e] = (px >= 0 && px < W) ? src[px] : (ushort)0;
            }
          }
        }
        *reinterpret_cast<s16x8*>(&dp[(r * 64 + co) * WGV2_DPW + blk * 8]) =
            *reinterpret_cast<const s16x8*>(vals);
      }
      // ---- stage the RB new X rows of the sliding window
#pragma unroll
      for (int r = 0; r < RB; ++r)
        stage_x_row(xb, KS == 3 ? oy0 + r + 1 : oy0 + r);
      __syncthreads();

      if (abl != 2) {
        // ---- compute: wave w covers co rows [co0+w*16, +16)
#pragma unroll
        for (int r = 0; r < RB; ++r) {
          const int oy = oy0 + r;
          const ushort* dprow = &dp[((r * 64) + wave * 16 + (lane & 15))
                                    * WGV2_DPW + 8 + kgrp];
          const s16x8 a_0 = *reinterpret_cast<const s16x8*>(dprow);
          s16x8 afrag[NTAP == 1 ? 1 : 3];
          if (KS == 1) {
            afrag[0] = a_0;
          } else {
            const s16x8 a_m = *reinterpret_cast<const s16x8*>(dprow - 8);
            const s16x8 a_p = *reinterpret_cast<const s16x8*>(dprow + 8);
            // dp index = u + PAD - kx: kx=0 -> +1, kx=1 -> 0, kx=2 -> -1
            afrag[0] =
                __builtin_shufflevector(a_0, a_p, 1, 2, 3, 4, 5, 6, 7, 8);
            afrag[1] = a_0;
            afrag[2] =
                __builtin_shufflevector(a_m, a_0, 7, 8, 9, 10, 11, 12, 13, 14);
          }
#pragma unroll
          for (int ky = 0; ky < KS; ++ky) {
            const int slot = KS == 1 ? oy % NSLOT
                : (((oy + ky - PAD) % NSLOT) + NSLOT) % NSLOT;
#pragma unroll
            for (int nci = 0; nci < 2; ++nci) {
              const s16x8 bfrag = *reinterpret_cast<const s16x8*>(
                  &xs[((nci * 16 + (lane & 15)) * NSLOT + slot) * WGV2_XW
                      + kgrp]);
#pragma unroll
              for (int kx = 0; kx < KS; ++kx)
                acc[ky * KS + kx][nci] =
                    __builtin_amdgcn_mfma_f32_16x16x32_bf16(
                        afrag[kx], bfrag, acc[ky * KS + kx][nci], 0, 0, 0);
            }
          }
        }
      }
      __syncthreads();
    }
  }

  // ---- flush: plain coalesced stores of this block's partial tile into
  // scratch [g][32ci][64co][NTAP]; a second kernel reduces over g.  No
  // atomics (an atomic flood here measured 70-87% of the kernel) and the
  // reduction order is DETERMINISTIC.
  if (abl == 1) return;
  float* part = dwp + ((((long long)blockIdx.x * gridDim.y + blockIdx.y)
                        * gridDim.z + blockIdx.z) * (32 * 64 * NTAP));
#pragma unroll
  for (int tap = 0; tap < NTAP; ++tap) {
#pragma unroll
    for (int nci = 0; nci < 2; ++nci) {
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int co = wave * 16 + (lane >> 4) * 4 + j;
        const int ci = nci * 16 + (lane & 15);
        part[((long long)ci * 64 + co) * NTAP + tap] = acc[tap][nci][j];
      }
    }
  }
}

// reduce the per-block partials: dwp[ci][co][tap] = sum_g part[g][...]
template <int NTAP>
__global__ void conv2d_wgrad_reduce_kernel(
    long long n, const float* __restrict__ part, int ng, int ciblks,
    int coblks, int Cin_p, int Cout_p, float* __restrict__ dwp) {
  ESR_KERNEL_LOOP(i, n) {  // i -> (ci, co, tap) over the PADDED dW
    const int tap = (int)(i % NTAP);
    const int co = (int)((i / NTAP) % Cout_p);
    const int ci = (int)(i / ((long long)NTAP * Cout_p));
    const int cib = ci / 32, cob = co / 64;
    const long long base =
        ((long long)cib * coblks + cob) * (32 * 64 * NTAP)
        + (((long long)(ci % 32) * 64 + (co % 64)) * NTAP + tap);
    const long long gstride = (long long)ciblks * coblks * (32 * 64 * NTAP);
    float v = 0.f;
    for (int g = 0; g < ng; ++g) v += part[g * gstride + base];
    dwp[i] = v;
  }
}

// ---------------------------------------------------------------------------
// Activation backward: dpre = dy * act'(y) elementwise, bf16.
// ---------------------------------------------------------------------------

template <int ACT>
__global__ void act_grad_kernel(long long n, const ushort* __restrict__ dy,
                                const ushort* __restrict__ y,
                                ushort* __restrict__ dpre) {
  ESR_KERNEL_LOOP(i, n) {
    const float g = bf16u_to_f(dy[i]);
    const float v = bf16u_to_f(y[i]);
    float d;
    if (ACT == 1) d = v > 0.f ? g : 0.f;
    else if (ACT == 2) d = g * v * (1.f - v);
    else d = g * (1.f - v * v);  // tanh
    dpre[i] = f_to_bf16u(d);
  }
}

// ---------------------------------------------------------------------------
// 16x16x32 GEMM probe: C[16,16] = A[16,32] @ B[32,16] — verifies the MFMA
// fragment maps this file assumes (guide §3; asymmetric-B testable).
// ---------------------------------------------------------------------------

__global__ void gemm16_probe_kernel(const ushort* __restrict__ A,
                                    const ushort* __restrict__ B,
                                    float* __restrict__ C) {
  const int lane = threadIdx.x & 63;
  s16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = (short)A[(lane & 15) * 32 + (lane >> 4) * 8 + j];
    b[j] = (short)B[((lane >> 4) * 8 + j) * 16 + (lane & 15)];
  }
  f32x4 acc = {};
  acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
#pragma unroll
  for (int j = 0; j < 4; ++j)
    C[((lane >> 4) * 4 + j) * 16 + (lane & 15)] = acc[j];
}

}  // namespace

// ===========================================================================
// Host wrappers
// ===========================================================================

#define DISPATCH_ACT(ACT, ...)                              \
  [&] {                                                     \
    switch (ACT) {                                          \
      case 0: { constexpr int kAct = 0; return __VA_ARGS__(); } \
      case 1: { constexpr int kAct = 1; return __VA_ARGS__(); } \
      case 2: { constexpr int kAct = 2; return __VA_ARGS__(); } \
      default: { constexpr int kAct = 3; return __VA_ARGS__(); } \
    }                                                       \
  }()

#define DISPATCH_KS_STRIDE(KS, STRIDE, ...)                                  \
  [&] {                                                                      \
    if (KS == 3 && STRIDE == 1) { constexpr int kKS = 3, kST = 1; return __VA_ARGS__(); } \
    if (KS == 3 && STRIDE == 2) { constexpr int kKS = 3, kST = 2; return __VA_ARGS__(); } \
    if (KS == 1 && STRIDE == 1) { constexpr int kKS = 1, kST = 1; return __VA_ARGS__(); } \
    TORCH_CHECK(false, "unsupported (ks, stride) = (", KS, ", ", STRIDE, ")"); \
    constexpr int kKS = 3, kST = 1; return __VA_ARGS__();                    \
  }()

static void check_bf16_4d(const at::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda() && t.scalar_type() == at::kBFloat16 &&
              t.is_contiguous(), name, ": contiguous bf16 CUDA tensor required");
}

// x [B,Cin,H,W] bf16, wp [ntap, Cout_p, Cin_p] bf16, bias [Cout] fp32 | none
at::Tensor conv2d_fwd_mfma(const at::Tensor& x, const at::Tensor& wp,
                           const c10::optional<at::Tensor>& bias,
                           int64_t Cout, int64_t ks, int64_t stride,
                           int64_t act) {
  check_bf16_4d(x, "conv2d_fwd_mfma: x");
  TORCH_CHECK(wp.is_cuda() && wp.scalar_type() == at::kBFloat16 &&
              wp.is_contiguous() && wp.dim() == 3, "wp: packed bf16 [ntap,Cop,Cip]");
  const int B = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout_p = wp.size(1), Cin_p = wp.size(2);
  TORCH_CHECK(wp.size(0) == ks * ks && Cin_p % CIK == 0 && Cout_p % 16 == 0);
  const int Ho = (H + 2 * (int)(ks / 2) - (int)ks) / (int)stride + 1;
  const int Wo = (W + 2 * (int)(ks / 2) - (int)ks) / (int)stride + 1;
  auto y = at::empty({B, Cout, Ho, Wo}, x.options());
  const float* bias_ptr = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->scalar_type() == at::kFloat && bias->is_contiguous());
    bias_ptr = bias->data_ptr<float>();
  }
  const int mrep = (Cout > 32 && Cout_p % 32 == 0) ? 2 : 1;
  const int ntx = (Wo + TW - 1) / TW, nty = (Ho + TH - 1) / TH;
  dim3 grid(ntx * nty, B, (Cout_p + 32 * mrep - 1) / (32 * mrep));
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_KS_STRIDE((int)ks, (int)stride, [&] {
    DISPATCH_ACT((int)act, [&] {
      if (mrep == 2)
        hipLaunchKernelGGL((conv2d_fwd_mfma_kernel<kKS, kST, kAct, 2>),
                           grid, dim3(256), 0, stream,
                           (const ushort*)x.data_ptr(), (const ushort*)wp.data_ptr(),
                           bias_ptr, (ushort*)y.data_ptr(),
                           Cin, H, W, (int)Cout, Ho, Wo, Cin_p, Cout_p, ntx);
      else
        hipLaunchKernelGGL((conv2d_fwd_mfma_kernel<kKS, kST, kAct, 1>),
                           grid, dim3(256), 0, stream,
                           (const ushort*)x.data_ptr(), (const ushort*)wp.data_ptr(),
                           bias_ptr, (ushort*)y.data_ptr(),
                           Cin, H, W, (int)Cout, Ho, Wo, Cin_p, Cout_p, ntx);
      return 0;
    });
    return 0;
  });
  C10_HIP_KERNEL_LAUNCH_CHECK();
  return y;
}

// x [B,Cin,H,W] bf16, w [Cout,Cin,ks,ks] bf16 (unpacked), bias fp32 | none
at::Tensor conv2d_fwd_valu(const at::Tensor& x, const at::Tensor& w,
                           const c10::optional<at::Tensor>& bias,
                           int64_t stride, int64_t act) {
  check_bf16_4d(x, "conv2d_fwd_valu: x");
  check_bf16_4d(w, "conv2d_fwd_valu: w");
  const int B = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = w.size(0), ks = w.size(2);
  const int Ho = (H + 2 * (ks / 2) - ks) / (int)stride + 1;
  const int Wo = (W + 2 * (ks / 2) - ks) / (int)stride + 1;
  auto y = at::empty({B, Cout, Ho, Wo}, x.options());
  const float* bias_ptr = nullptr;
  if (bias.has_value()) bias_ptr = bias->data_ptr<float>();
  const long long n = (long long)B * Cout * Ho * Wo;
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_KS_STRIDE(ks, (int)stride, [&] {
    DISPATCH_ACT((int)act, [&] {
      hipLaunchKernelGGL((conv2d_fwd_valu_kernel<kKS, kST, kAct>),
                         dim3(esr_grid(n)), dim3(ESR_BLOCK), 0, stream,
                         n, (const ushort*)x.data_ptr(), (const ushort*)w.data_ptr(),
                         bias_ptr, (ushort*)y.data_ptr(), Cin, H, W, Cout, Ho, Wo);
      return 0;
    });
    return 0;
  });
  C10_HIP_KERNEL_LAUNCH_CHECK();
  return y;
}

// v2 strip kernel for tiny-channel shapes (see kernel comment)
at::Tensor conv2d_fwd_valu2(const at::Tensor& x, const at::Tensor& w,
                            const c10::optional<at::Tensor>& bias,
                            int64_t stride, int64_t act) {
  check_bf16_4d(x, "conv2d_fwd_valu2: x");
  check_bf16_4d(w, "conv2d_fwd_valu2: w");
  const int B = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = w.size(0), ks = w.size(2);
  TORCH_CHECK(Cin <= 32, "valu2 is for tiny-channel shapes (Cin <= 32)");
  const int Ho = (H + 2 * (ks / 2) - ks) / (int)stride + 1;
  const int Wo = (W + 2 * (ks / 2) - ks) / (int)stride + 1;
  auto y = at::empty({B, Cout, Ho, Wo}, x.options());
  const float* bias_ptr = nullptr;
  if (bias.has_value()) bias_ptr = bias->data_ptr<float>();
  const int coch = Cout >= 8 || Cout > 2 ? 8 : 2;
  const int nsx = (Wo + 7) / 8;
  const long long nstrips = (long long)B * Ho * nsx;
  dim3 grid(esr_grid(nstrips), 1, (Cout + coch - 1) / coch);
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_KS_STRIDE(ks, (int)stride, [&] {
    DISPATCH_ACT((int)act, [&] {
      if (coch == 8)
        hipLaunchKernelGGL((conv2d_fwd_valu2_kernel<kKS, kST, kAct, 8>),
                           grid, dim3(256), 0, stream,
                           (const ushort*)x.data_ptr(),
                           (const ushort*)w.data_ptr(), bias_ptr,
                           (ushort*)y.data_ptr(), Cin, H, W, Cout, Ho, Wo,
                           nstrips, nsx);
      else
        hipLaunchKernelGGL((conv2d_fwd_valu2_kernel<kKS, kST, kAct, 2>),
                           grid, dim3(256), 0, stream,
                           (const ushort*)x.data_ptr(),
                           (const ushort*)w.data_ptr(), bias_ptr,
                           (ushort*)y.data_ptr(), Cin, H, W, Cout, Ho, Wo,
                           nstrips, nsx);
      return 0;
    });
    return 0;
  });
  C10_HIP_KERNEL_LAUNCH_CHECK();
  return y;
}

at::Tensor conv2d_dgrad_s2(const at::Tensor& dy, const at::Tensor& w,
                           int64_t H, int64_t W) {
  check_bf16_4d(dy, "conv2d_dgrad_s2: dy");
  check_bf16_4d(w, "conv2d_dgrad_s2: w");
  const int B = dy.size(0), Cout = dy.size(1), Ho = dy.size(2), Wo = dy.size(3);
  const int Cin = w.size(1), ks = w.size(2);
  TORCH_CHECK(ks == 3, "dgrad_s2 supports 3x3 only");
  auto dx = at::empty({B, Cin, H, W}, dy.options());
  const long long n = (long long)B * Cin * H * W;
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((conv2d_dgrad_s2_valu_kernel<3>),
                     dim3(esr_grid(n)), dim3(ESR_BLOCK), 0, stream,
                     n, (const ushort*)dy.data_ptr(), (const ushort*)w.data_ptr(),
                     (ushort*)dx.data_ptr(), Cin, (int)H, (int)W, Cout, Ho, Wo);
  C10_HIP_KERNEL_LAUNCH_CHECK();
  return dx;
}

// returns padded fp32 dW [Cout_p, Cin_p, ks, ks]
at::Tensor conv2d_wgrad_mfma(const at::Tensor& x, const at::Tensor& dpre,
                             int64_t ks, int64_t stride,
                             int64_t Cin_p, int64_t Cout_p) {
  check_bf16_4d(x, "conv2d_wgrad: x");
  check_bf16_4d(dpre, "conv2d_wgrad: dpre");
  const int B = x.size(0), Cin = x.size(1), H = x.size(2), W = x.size(3);
  const int Cout = dpre.size(1), Ho = dpre.size(2), Wo = dpre.size(3);
  TORCH_CHECK(Cin_p % 16 == 0 && Cout_p % 16 == 0);
  // TRANSPOSED result [Cin_p, Cout_p, ks, ks]; the python layer permutes
  // back when slicing
  auto dwp = at::zeros({Cin_p, Cout_p, ks, ks},
                       x.options().dtype(at::kFloat));
  const int uw = (W + TW - 1) / TW;
  auto stream = at::hip::getCurrentHIPStream();
  if (stride == 1) {
    const int ciblks = (Cin_p + 31) / 32, coblks = (Cout_p + 63) / 64;
    // frames-per-block: accumulate several frames in registers before the
    // atomic flush (atomic traffic / fpb) while keeping the grid >= ~2048
    long long nb = (long long)B * uw * ciblks * coblks;
    int fpb = 1;
    while (fpb < 32 && fpb * 2 <= B && nb / (fpb * 2) >= 1024) fpb *= 2;
    static const int abl = [] {
      const char* e = getenv("ESR_WGRAD_ABL");
      return e ? atoi(e) : 0;
    }();
    const int bblks = (B + fpb - 1) / fpb;
    const int ng = bblks * uw;
    const int ntap = (int)(ks * ks);
    dim3 grid((unsigned)ng, ciblks, coblks);
    auto part = at::empty({(long long)ng * ciblks * coblks * 32 * 64 * ntap},
                          x.options().dtype(at::kFloat));
    if (ks == 3)
      hipLaunchKernelGGL((conv2d_wgrad_s1_kernel<3>), grid, dim3(256), 0,
                         stream, (const ushort*)x.data_ptr(),
                         (const ushort*)dpre.data_ptr(),
                         part.data_ptr<float>(),
                         Cin, H, W, Cout, (int)Cin_p, (int)Cout_p, uw, B,
                         fpb, abl);
    else
      hipLaunchKernelGGL((conv2d_wgrad_s1_kernel<1>), grid, dim3(256), 0,
                         stream, (const ushort*)x.data_ptr(),
                         (const ushort*)dpre.data_ptr(),
                         part.data_ptr<float>(),
                         Cin, H, W, Cout, (int)Cin_p, (int)Cout_p, uw, B,
                         fpb, abl);
    const long long nred = (long long)Cin_p * Cout_p * ntap;
    if (ks == 3)
      hipLaunchKernelGGL((conv2d_wgrad_reduce_kernel<9>),
                         dim3(esr_grid(nred)), dim3(ESR_BLOCK), 0, stream,
                         nred, part.data_ptr<float>(), ng, ciblks, coblks,
                         (int)Cin_p, (int)Cout_p, dwp.data_ptr<float>());
    else
      hipLaunchKernelGGL((conv2d_wgrad_reduce_kernel<1>),
                         dim3(esr_grid(nred)), dim3(ESR_BLOCK), 0, stream,
                         nred, part.data_ptr<float>(), ng, ciblks, coblks,
                         (int)Cin_p, (int)Cout_p, dwp.data_ptr<float>());
  } else {
    const int rows_per_blk = 16;
    const long long nslab =
        (long long)B * ((Ho + rows_per_blk - 1) / rows_per_blk) * uw;
    dim3 grid((unsigned)nslab, Cin_p / 16, Cout_p / 16);
    TORCH_CHECK(ks == 3, "stride-2 wgrad supports 3x3 only");
    hipLaunchKernelGGL((conv2d_wgrad_mfma_kernel<3, 2>),
                       grid, dim3(256), 0, stream,
                       (const ushort*)x.data_ptr(), (const ushort*)dpre.data_ptr(),
                       dwp.data_ptr<float>(), Cin, H, W, Cout, Ho, Wo,
                       (int)Cin_p, (int)Cout_p, rows_per_blk);
  }
  C10_HIP_KERNEL_LAUNCH_CHECK();
  return dwp;
}

at::Tensor act_grad(const at::Tensor& dy, const at::Tensor& y, int64_t act) {
  check_bf16_4d(dy, "act_grad: dy");
  auto dpre = at::empty_like(dy);
  const long long n = dy.numel();
  auto stream = at::hip::getCurrentHIPStream();
  DISPATCH_ACT((int)act, [&] {
    if (kAct == 0) { dpre.copy_(dy); return 0; }
    hipLaunchKernelGGL((act_grad_kernel<kAct>),
                       dim3(esr_grid(n)), dim3(ESR_BLOCK), 0, stream,
                       n, (const ushort*)dy.data_ptr(),
                       (const ushort*)y.data_ptr(), (ushort*)dpre.data_ptr());
    return 0;
  });
  C10_HIP_KERNEL_LAUNCH_CHECK();
  return dpre;
}

at::Tensor gemm16_probe(const at::Tensor& A, const at::Tensor& B) {
  TORCH_CHECK(A.is_cuda() && A.scalar_type() == at::kBFloat16 &&
              A.is_contiguous() && B.is_contiguous() &&
              A.sizes() == at::IntArrayRef({16, 32}) &&
              B.sizes() == at::IntArrayRef({32, 16}), "probe: A[16,32] B[32,16] bf16");
  auto C = at::empty({16, 16}, A.options().dtype(at::kFloat));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(gemm16_probe_kernel, dim3(1), dim3(64), 0, stream,
                     (const ushort*)A.data_ptr(), (const ushort*)B.data_ptr(),
                     C.data_ptr<float>());
  C10_HIP_KERNEL_LAUNCH_CHECK();
  return C;
}
