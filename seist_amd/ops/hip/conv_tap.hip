// Tap-gather MFMA conv1d forward / input-gradient for stride-1 bf16 convs
// (dense, grouped, dilated) — K3/K4/K6 of SURVEY.md §2.4 on the matrix
// cores.
//
//   fwd: y[n,co,l]  = sum_{ci in group, k} w[co,cig,k] * x[n,ci,l + k*d - padl]
//   dx:  dx[n,ci,l] = sum_{co in group, k} w[co,cig,k] * dy[n,co,l + padl - k*d]
//
// Unlike the im2col-in-LDS kernel (pw_mfma.hip conv_mfma_kernel), each input
// channel row is staged in LDS ONCE per (l-tile, ci-chunk) — the K
// tap-shifted B fragments are gathered from that single image, so HBM
// traffic is not multiplied by K. The GEMM reduce dimension runs over
// channel chunks of 32 with an inner tap loop; the A fragment for tap k is
// one contiguous bf16x8 from a [16 m][K][32 c] LDS weight image.
//
// Tile: 16 out-channels x 256 l per block (4 waves side by side along l,
// 4 accumulators each). Groups: the reduce-channel window is the group's
// channels when Cog >= 16 (tiles never straddle groups, Cog % 16 == 0);
// when Cog < 16 (several groups inside one 16-channel tile, Ci == Co,
// Cog == Cig) the window is the tile's own 16-channel span and off-group
// weights are staged as zeros — the same diagonal-tile trick as
// dw_mfma.hip, costing only MFMA issues, never bandwidth.
//
// Fragment maps as in pw_mfma.hip (gfx950 v_mfma_f32_16x16x32_bf16):
// A[i=lane&15][k=(lane>>4)*8+j], B[k=(lane>>4)*8+j][n=lane&15],
// D col=lane&15, row=(lane>>4)*4+reg.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

typedef __bf16 sa_bf16;
typedef sa_bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

namespace {

constexpr int kBlock = 256;
constexpr int kLT = 256;   // l per block (4 waves x 64)
constexpr int kCT = 32;    // reduce-channel chunk per K-step
constexpr int kMaxLds = 56 * 1024;

template <bool IS_DX, bool HAS_BIAS>
__global__ __launch_bounds__(kBlock)
void conv_tap_kernel(const sa_bf16* __restrict__ x,
                     const sa_bf16* __restrict__ w,
                     const sa_bf16* __restrict__ bias,
                     sa_bf16* __restrict__ y,
                     float* __restrict__ stats,
                     int N, int Cin, int Cout, long Lin, long Lout,
                     int K, int padl, int dil, int G, int xext,
                     int xpitch) {
  extern __shared__ sa_bf16 smem[];
  sa_bf16* w_s = smem;                    // [16][K][kCT] (c contiguous)
  sa_bf16* x_s = smem + 16 * K * kCT;     // [kCT][xpitch]
  __shared__ float stats_s[16 * 2];

  const int n = blockIdx.y;
  const int Cg_out = Cout / G;  // out channels per group (m axis)
  const int Cg_in = Cin / G;    // reduce channels per group
  int m0;   // out-channel tile start (global)
  int w0, wl;  // reduce-channel window start (global) and length
  if (G == 1 || Cg_out >= 16) {
    const int tpg = (G == 1) ? (Cout + 15) / 16 : Cg_out / 16;
    const int g = blockIdx.z / tpg;
    m0 = g * Cg_out + (blockIdx.z - g * tpg) * 16;
    w0 = g * Cg_in;
    wl = Cg_in;
  } else {
    // several groups per tile (Cin == Cout, Cg_out == Cg_in): the reduce
    // window is the tile's own channel span; off-group weights stage as 0
    m0 = blockIdx.z * 16;
    w0 = m0;
    wl = min(16, Cin - w0);
  }
  const long l0 = (long)blockIdx.x * kLT;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int frag_m = lane & 15;
  const int kbase = (lane >> 4) * 8;

  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc2 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc3 = {0.f, 0.f, 0.f, 0.f};

  // stage start: covers every tap shift for l in [l0, l0+kLT)
  const long min_off = IS_DX ? (long)padl - (long)(K - 1) * dil : -(long)padl;
  const long s0 = (l0 + min_off) & ~7L;
  const int base = (int)(l0 + min_off - s0);  // in [0, 8)

  const sa_bf16* xb = x + (long)n * Cin * Lin;

  for (int c0 = 0; c0 < wl; c0 += kCT) {
    const int cn = min(kCT, wl - c0);
    __syncthreads();
    // ---- weight chunk [16 m][K][kCT c], zero off-range / off-group ----
    for (int idx = tid; idx < 16 * K * kCT; idx += kBlock) {
      const int m = idx / (K * kCT);
      const int r = idx - m * K * kCT;
      const int k = r / kCT;
      const int c = r - k * kCT;
      const int mg = m0 + m;                 // output-tensor channel
      const int cg = w0 + c0 + c;            // reduce (input-tensor) channel
      float v = 0.0f;
      if (mg < Cout && c < cn) {
        // conv-space roles: dx swaps them (x param is dy, y param is dx)
        const int convCo = IS_DX ? Cin : Cout;
        const int convCi = IS_DX ? Cout : Cin;
        const int co = IS_DX ? cg : mg;
        const int ci = IS_DX ? mg : cg;
        const int gg = co / (convCo / G);
        if (ci / (convCi / G) == gg) {       // same group, else zero
          const int cig = ci - gg * (convCi / G);
          v = (float)w[((long)co * (convCi / G) + cig) * K + k];
        }
      }
      w_s[(m * K + k) * kCT + c] = (sa_bf16)v;
    }
    // ---- input rows [kCT c][xext l], staged once (aligned b128) ----
    for (int idx = tid; idx < kCT * (xext / 8); idx += kBlock) {
      const int c = idx / (xext / 8);
      const int e8 = idx - c * (xext / 8);
      const long gl = s0 + e8 * 8;
      bf16x8 v = {};
      if (c < cn) {
        const sa_bf16* row = xb + (long)(w0 + c0 + c) * Lin;
        if (gl >= 0 && gl + 8 <= Lin) {
          v = *(const bf16x8*)(row + gl);
        } else {
          for (int j = 0; j < 8; ++j) {
            const long lj = gl + j;
            if (lj >= 0 && lj < Lin) v[j] = row[lj];
          }
        }
      }
      *(bf16x8*)(x_s + c * xpitch + e8 * 8) = v;
    }
    __syncthreads();

    const int lb = wid * 64 + frag_m;  // this lane's l column within tile
    for (int k = 0; k < K; ++k) {
      const bf16x8 a =
          *(const bf16x8*)(w_s + (frag_m * K + k) * kCT + kbase);
      // tap offset relative to min_off: fwd k*d, dx (K-1-k)*d
      const int toff = base + (IS_DX ? (K - 1 - k) * dil : k * dil);
#pragma unroll
      for (int nrep = 0; nrep < 4; ++nrep) {
        bf16x8 b;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          b[j] = x_s[(kbase + j) * xpitch + toff + nrep * 16 + lb];
        }
        switch (nrep) {
          case 0: acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc0, 0, 0, 0); break;
          case 1: acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc1, 0, 0, 0); break;
          case 2: acc2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc2, 0, 0, 0); break;
          case 3: acc3 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc3, 0, 0, 0); break;
        }
      }
    }
  }

  // ---- epilogue: D col = l, row = out channel ----
  const int d_col = lane & 15;
  const int d_row0 = (lane >> 4) * 4;
  const int Cm = Cout;
  float ssum[4] = {0.f, 0.f, 0.f, 0.f};
  float ssum2[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int nrep = 0; nrep < 4; ++nrep) {
    const f32x4 acc = (nrep == 0) ? acc0 : (nrep == 1) ? acc1
                      : (nrep == 2) ? acc2 : acc3;
    const long lg = l0 + wid * 64 + nrep * 16 + d_col;
    if (lg >= Lout) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int mg = m0 + d_row0 + r;
      if (mg < Cm) {
        float v = acc[r];
        if (HAS_BIAS) v += (float)bias[mg];
        const sa_bf16 vb = (sa_bf16)v;
        y[((long)n * Cm + mg) * Lout + lg] = vb;
        if (stats != nullptr) {
          const float vf = (float)vb;
          ssum[r] += vf;
          ssum2[r] += vf * vf;
        }
      }
    }
  }
  if (stats != nullptr) {
    __syncthreads();
    for (int t = tid; t < 16 * 2; t += kBlock) stats_s[t] = 0.0f;
    __syncthreads();
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int b = 1; b < 16; b <<= 1) {
        ssum[r] += __shfl_xor(ssum[r], b, sa::kWave);
        ssum2[r] += __shfl_xor(ssum2[r], b, sa::kWave);
      }
      if (d_col == 0) {
        atomicAdd(&stats_s[(d_row0 + r) * 2 + 0], ssum[r]);
        atomicAdd(&stats_s[(d_row0 + r) * 2 + 1], ssum2[r]);
      }
    }
    __syncthreads();
    const long split = (long)blockIdx.y * gridDim.x + blockIdx.x;
    for (int t = tid; t < 16 * 2; t += kBlock) {
      const int mg = m0 + (t >> 1);
      if (mg < Cm) {
        stats[(split * Cm + mg) * 2 + (t & 1)] = stats_s[t];
      }
    }
  }
}

}  // namespace

// returns false if the shape/dtype is outside this kernel's envelope
bool conv_tap_mfma(const at::Tensor& x, const at::Tensor& w,
                   const c10::optional<at::Tensor>& bias, at::Tensor& y,
                   long padl, long dilation, long groups, bool is_dx,
                   at::Tensor* stats_out) {
  if (x.scalar_type() != at::kBFloat16 || w.scalar_type() != at::kBFloat16)
    return false;
  const int N = x.size(0), Cin = x.size(1);
  const long Lin = x.size(2);
  const int Cout = y.size(1);
  const long Lout = y.size(2);
  const int K = w.size(2);
  const int G = (int)groups;
  const int Cg_out = Cout / G;
  const int Cg_in = Cin / G;
  if (K < 1 || K > 24) return false;
  static const bool no_tap = getenv("SEIST_AMD_NO_TAP") != nullptr;
  if (no_tap && G == 1) return false;  // A/B: dense goes to conv_mfma/VALU
  // measured: at reduce-width < 8 channels the im2col-in-LDS kernel wins
  // (e.g. Ci=3,K=11,L=8192: 121 vs 211 us) — let it take those
  if (G == 1 && Cin < 8) return false;
  if (G > 1) {
    // grouped tiles need group-aligned 16-channel windows
    if (Cg_out != Cg_in || Cin != Cout) return false;
    if (Cg_out >= 16 ? (Cg_out % 16 != 0) : (16 % Cg_out != 0)) return false;
    if (Cg_out < 16 && Cg_out == 1) return false;  // depthwise: direct kernel
  }
  const int xext = (kLT + (K - 1) * (int)dilation + 8 + 7) & ~7;
  const int xpitch = xext + 3;
  const size_t lds = sizeof(sa_bf16) * (16 * K * kCT + kCT * xpitch);
  if (lds > kMaxLds) return false;

  const int mtiles = (G == 1) ? (Cout + 15) / 16
                              : (Cg_out >= 16 ? (Cg_out / 16) * G
                                              : (Cout + 15) / 16);
  dim3 grid(sa::ceil_div(Lout, (long)kLT), N, mtiles);
  auto stream = at::hip::getCurrentHIPStream();
  const bool has_bias = bias.has_value() && bias->defined();
  at::Tensor bct;
  if (has_bias) bct = bias->to(x.scalar_type()).contiguous();
  const sa_bf16* bp =
      has_bias ? (const sa_bf16*)bct.data_ptr() : nullptr;

  float* sp = nullptr;
  if (stats_out != nullptr) {
    const long nsplit = (long)N * grid.x;
    *stats_out = at::empty({nsplit, Cout, 2}, x.options().dtype(at::kFloat));
    sp = stats_out->data_ptr<float>();
  }

  auto launch = [&](auto dx_t, auto hb_t) {
    hipLaunchKernelGGL(
        (conv_tap_kernel<decltype(dx_t)::value, decltype(hb_t)::value>),
        grid, dim3(kBlock), lds, stream.stream(),
        (const sa_bf16*)x.data_ptr(), (const sa_bf16*)w.data_ptr(), bp,
        (sa_bf16*)y.data_ptr(), sp, N, Cin, Cout, Lin, Lout, K, (int)padl,
        (int)dilation, G, xext, xpitch);
  };
  if (is_dx) {
    if (has_bias) launch(std::true_type{}, std::true_type{});
    else launch(std::true_type{}, std::false_type{});
  } else {
    if (has_bias) launch(std::false_type{}, std::true_type{});
    else launch(std::false_type{}, std::false_type{});
  }
  return true;
}

// ---------------------------------------------------------------------------
// Strided variant (S in {2, 4}, dense, dilation 1): PhaseNet's stride-4
// encoder convs and the transposed-conv gather. 64-wide l tile (the x
// window is LT*S + K wide, so the 256-l tile of the stride-1 kernel would
// not fit in LDS at S=4); 4 waves side by side along l, one 16x16 output
// fragment each. Same staging discipline: each input channel row staged
// once per (l-tile, ci-chunk); B fragments gathered at stride S (fwd) or
// with the % S divisibility mask (dx).
// ---------------------------------------------------------------------------

namespace {

constexpr int kLTS = 64;  // l per block in the strided kernel

template <bool IS_DX, bool HAS_BIAS, int S>
__global__ __launch_bounds__(kBlock)
void conv_tap_s_kernel(const sa_bf16* __restrict__ x,
                       const sa_bf16* __restrict__ w,
                       const sa_bf16* __restrict__ bias,
                       sa_bf16* __restrict__ y,
                       float* __restrict__ stats,
                       int N, int Cin, int Cout, long Lin, long Lout,
                       int K, int padl, int xext, int xpitch) {
  extern __shared__ sa_bf16 smem[];
  sa_bf16* w_s = smem;                         // [16][K][kCT] (c contiguous)
  sa_bf16* x_s = smem + 16 * K * kCT;          // [kCT][xpitch]
  __shared__ float stats_s[16 * 2];

  const int n = blockIdx.y;
  const int m0 = blockIdx.z * 16;
  const long l0 = (long)blockIdx.x * kLTS;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int frag_m = lane & 15;
  const int kbase = (lane >> 4) * 8;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};

  // fwd: x index = l*S - padl + k; staged from aligned s0
  // dx:  dy index = (l + padl - k)/S where divisible; staged likewise
  long s0;
  int base;  // l0's first index relative to s0
  long lob = 0;
  if (IS_DX) {
    lob = (l0 + padl - (long)(K - 1)) / (long)S - 1;
    if (lob < 0) lob = 0;
    s0 = 0;  // unused in dx mode
    base = 0;
  } else {
    s0 = (l0 * S - padl) & ~7L;
    base = (int)(l0 * S - padl - s0);
  }

  const sa_bf16* xb = x + (long)n * Cin * Lin;

  for (int c0 = 0; c0 < Cin; c0 += kCT) {
    const int cn = min(kCT, Cin - c0);
    __syncthreads();
    // ---- weight chunk [16 m][K][kCT c] ----
    for (int idx = tid; idx < 16 * K * kCT; idx += kBlock) {
      const int m = idx / (K * kCT);
      const int r = idx - m * K * kCT;
      const int k = r / kCT;
      const int c = r - k * kCT;
      const int mg = m0 + m;
      const int cg = c0 + c;
      float v = 0.0f;
      if (mg < Cout && c < cn) {
        const int co = IS_DX ? cg : mg;
        const int ci = IS_DX ? mg : cg;
        v = (float)w[((long)co * (IS_DX ? Cout : Cin) + ci) * K + k];
      }
      w_s[(m * K + k) * kCT + c] = (sa_bf16)v;
    }
    // ---- input rows [kCT c][xext], staged once (aligned b128) ----
    for (int idx = tid; idx < kCT * (xext / 8); idx += kBlock) {
      const int c = idx / (xext / 8);
      const int e8 = idx - c * (xext / 8);
      const long gl = (IS_DX ? lob : s0) + e8 * 8;
      bf16x8 v = {};
      if (c < cn) {
        const sa_bf16* row = xb + (long)(c0 + c) * Lin;
        if (gl >= 0 && gl + 8 <= Lin) {
          v = *(const bf16x8*)(row + gl);
        } else {
          for (int j = 0; j < 8; ++j) {
            const long lj = gl + j;
            if (lj >= 0 && lj < Lin) v[j] = row[lj];
          }
        }
      }
      *(bf16x8*)(x_s + c * xpitch + e8 * 8) = v;
    }
    __syncthreads();

    const int ll = wid * 16 + (lane & 15);  // this lane's l column in tile
    for (int k = 0; k < K; ++k) {
      const bf16x8 a =
          *(const bf16x8*)(w_s + (frag_m * K + k) * kCT + kbase);
      int idxl = -1;
      if (IS_DX) {
        const long num = l0 + ll + padl - k;
        if (num >= 0 && num % S == 0) {
          const long lo = num / S;
          if (lo < Lin) idxl = (int)(lo - lob);
        }
      } else {
        idxl = base + ll * S + k;
      }
      bf16x8 b = {};
      if (idxl >= 0) {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          b[j] = x_s[(kbase + j) * xpitch + idxl];
        }
      }
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
    }
  }

  // ---- epilogue: D col = l, row = out channel ----
  const int d_col = lane & 15;
  const int d_row0 = (lane >> 4) * 4;
  const long lg = l0 + wid * 16 + d_col;
  float ssum[4] = {0.f, 0.f, 0.f, 0.f};
  float ssum2[4] = {0.f, 0.f, 0.f, 0.f};
  if (lg < Lout) {
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int mg = m0 + d_row0 + r;
      if (mg < Cout) {
        float v = acc[r];
        if (HAS_BIAS) v += (float)bias[mg];
        const sa_bf16 vb = (sa_bf16)v;
        y[((long)n * Cout + mg) * Lout + lg] = vb;
        if (stats != nullptr) {
          const float vf = (float)vb;
          ssum[r] += vf;
          ssum2[r] += vf * vf;
        }
      }
    }
  }
  if (stats != nullptr) {
    __syncthreads();
    for (int t = tid; t < 16 * 2; t += kBlock) stats_s[t] = 0.0f;
    __syncthreads();
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int b = 1; b < 16; b <<= 1) {
        ssum[r] += __shfl_xor(ssum[r], b, sa::kWave);
        ssum2[r] += __shfl_xor(ssum2[r], b, sa::kWave);
      }
      if (d_col == 0) {
        atomicAdd(&stats_s[(d_row0 + r) * 2 + 0], ssum[r]);
        atomicAdd(&stats_s[(d_row0 + r) * 2 + 1], ssum2[r]);
      }
    }
    __syncthreads();
    const long split = (long)blockIdx.y * gridDim.x + blockIdx.x;
    for (int t = tid; t < 16 * 2; t += kBlock) {
      const int mg = m0 + (t >> 1);
      if (mg < Cout) {
        stats[(split * Cout + mg) * 2 + (t & 1)] = stats_s[t];
      }
    }
  }
}

}  // namespace

// strided dense path; returns false outside the envelope
bool conv_tap_s_mfma(const at::Tensor& x, const at::Tensor& w,
                     const c10::optional<at::Tensor>& bias, at::Tensor& y,
                     long stride, long padl, long dilation, long groups,
                     bool is_dx, at::Tensor* stats_out) {
  if (x.scalar_type() != at::kBFloat16 || w.scalar_type() != at::kBFloat16)
    return false;
  if (groups != 1 || dilation != 1 || (stride != 2 && stride != 4))
    return false;
  const int N = x.size(0), Cin = x.size(1);
  const long Lin = x.size(2);
  const int Cout = y.size(1);
  const long Lout = y.size(2);
  const int K = w.size(2);
  if (K < 1 || K > 24) return false;
  // fwd window: kLTS*S + K + slack; dx window: kLTS/S + K/S + slack
  const int xext = is_dx
      ? ((kLTS / (int)stride + K / (int)stride + 12 + 7) & ~7)
      : ((kLTS * (int)stride + K + 16 + 7) & ~7);
  const int xpitch = xext + 3;
  const size_t lds = sizeof(sa_bf16) * (16 * K * kCT + kCT * xpitch);
  if (lds > kMaxLds) return false;

  dim3 grid(sa::ceil_div(Lout, (long)kLTS), N, (Cout + 15) / 16);
  auto stream = at::hip::getCurrentHIPStream();
  const bool has_bias = bias.has_value() && bias->defined();
  at::Tensor bct;
  if (has_bias) bct = bias->to(x.scalar_type()).contiguous();
  const sa_bf16* bp = has_bias ? (const sa_bf16*)bct.data_ptr() : nullptr;

  float* sp = nullptr;
  if (stats_out != nullptr) {
    const long nsplit = (long)N * grid.x;
    *stats_out = at::empty({nsplit, Cout, 2}, x.options().dtype(at::kFloat));
    sp = stats_out->data_ptr<float>();
  }

  auto launch = [&](auto dx_t, auto hb_t, auto s_t) {
    hipLaunchKernelGGL(
        (conv_tap_s_kernel<decltype(dx_t)::value, decltype(hb_t)::value,
                           decltype(s_t)::value>),
        grid, dim3(kBlock), lds, stream.stream(),
        (const sa_bf16*)x.data_ptr(), (const sa_bf16*)w.data_ptr(), bp,
        (sa_bf16*)y.data_ptr(), sp, N, Cin, Cout, Lin, Lout, K, (int)padl,
        xext, xpitch);
  };
  auto launch_hb = [&](auto dx_t, auto s_t) {
    if (has_bias) launch(dx_t, std::true_type{}, s_t);
    else launch(dx_t, std::false_type{}, s_t);
  };
  auto launch_dx = [&](auto s_t) {
    if (is_dx) launch_hb(std::true_type{}, s_t);
    else launch_hb(std::false_type{}, s_t);
  };
  if (stride == 2) launch_dx(std::integral_constant<int, 2>{});
  else launch_dx(std::integral_constant<int, 4>{});
  return true;
}
