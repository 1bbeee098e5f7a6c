// Direct Conv1d (dense / grouped / depthwise / dilated-causal) —
// K2/K3/K4/K6 of SURVEY.md §2.4. Covers the reference's depthwise stem
// convs (models/seist.py:134-141), grouped convs (:215-222), dense head /
// PhaseNet / EQT convs, and dist-PT's dilated causal convs
// (models/distpt_network.py:17-87).
//
// x (N, Ci, L), w (Co, Cig=Ci/G, K), y (N, Co, Lo)
// y[n][co][lo] = sum_{cig,k} w[co][cig][k] * x[n][g*Cig+cig][lo*s - padl + k*d]
//
// Two forward/dx structures:
//  * depthwise (Cog == 1): direct per-(n,co) kernel, weights in LDS, x
//    window through L1 — each x element is consumed by only one output
//    channel, so there is nothing to share.
//  * dense/grouped (Cog > 1): "conv-GEMM" — a 32-wide output-channel chunk
//    per block shares every staged x (or dy) read across 32 accumulators,
//    exactly like the pointwise GEMM kernel but with a tap loop. This cut
//    the head-conv forward ~10x vs the one-channel-per-block design
//    (rocprofv3, profiles/).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

at::Tensor sum_batch(const at::Tensor& in);
at::Tensor sum_mid(const at::Tensor& in);
at::Tensor sum_mid_to(const at::Tensor& in, at::ScalarType out_dtype);
at::Tensor sum_batch_to(const at::Tensor& in, at::ScalarType out_dtype);
at::Tensor conv_dw_smallc(const at::Tensor& dy, const at::Tensor& x,
                          long K, long padl, long stride, long dilation,
                          long groups, at::ScalarType out_dtype);
at::Tensor channel_sum_to(const at::Tensor& in, at::ScalarType out_dtype);
bool conv_mfma(const at::Tensor& x, const at::Tensor& w,
               const c10::optional<at::Tensor>& bias, at::Tensor& y,
               long padl, long dilation, bool is_dx,
               at::Tensor* stats_out = nullptr);
at::Tensor channel_sum(const at::Tensor& in);
at::Tensor bn_sums_only(const at::Tensor& x);
c10::optional<at::Tensor> dw_mfma_try(const at::Tensor& dy,
                                      const at::Tensor& x, long stride,
                                      long padl, long groups, long dilation,
                                      int K);
bool conv_tap_mfma(const at::Tensor& x, const at::Tensor& w,
                   const c10::optional<at::Tensor>& bias, at::Tensor& y,
                   long padl, long dilation, long groups, bool is_dx,
                   at::Tensor* stats_out = nullptr);
bool conv_smallc_mfma(const at::Tensor& x, const at::Tensor& w,
                      const c10::optional<at::Tensor>& bias, at::Tensor& y,
                      long padl, long dilation, long groups, bool is_dx,
                      at::Tensor* stats_out = nullptr);
bool conv_tap_s_mfma(const at::Tensor& x, const at::Tensor& w,
                     const c10::optional<at::Tensor>& bias, at::Tensor& y,
                     long stride, long padl, long dilation, long groups,
                     bool is_dx, at::Tensor* stats_out = nullptr);

namespace {

constexpr int kBlock = 256;
constexpr int kMaxWLds = 8192;  // floats of LDS weight stage per block
constexpr int kJT = 32;         // output-channel chunk of the conv-GEMM

// ---------------- depthwise direct ----------------
// One block per (256-wide lo chunk, n, channel). The input window the chunk
// touches (kBlock*stride + (K-1)*dil elements) is staged in LDS once with
// coalesced guarded loads, so the K-tap inner loop issues LDS reads instead
// of K global loads per output element (the global-load version was ~3x off
// roofline on the stems — VMEM instruction-issue bound, not bandwidth).
// 4 outputs per thread (1024-wide l tile): at 256-wide tiles the grid was
// ~128k sub-microsecond blocks and the step was dispatch-rate bound, not
// bandwidth bound.
constexpr int kDwTile = 4;

template <typename scalar_t, bool HAS_BIAS>
__global__ void dwconv_fwd_kernel(const scalar_t* __restrict__ x,
                                  const scalar_t* __restrict__ w,
                                  const scalar_t* __restrict__ bias,
                                  scalar_t* __restrict__ y,
                                  int N, int Ci, int Co, long L, long Lo,
                                  int K, int stride, int padl, int dil) {
  extern __shared__ float w_lds[];  // [K] floats, then the x window
  float* x_s = w_lds + K;
  const int n = blockIdx.y;
  const int co = blockIdx.z;
  const long lo0 = (long)blockIdx.x * (kBlock * kDwTile);
  const long s0 = lo0 * stride - padl;
  const int ext = kBlock * kDwTile * stride + (K - 1) * dil + 1;
  const scalar_t* xr = x + ((long)n * Ci + co) * L;  // depthwise: ci == co
  for (int idx = threadIdx.x; idx < K; idx += kBlock) {
    w_lds[idx] = (float)w[(long)co * K + idx];
  }
  for (int idx = threadIdx.x; idx < ext; idx += kBlock) {
    const long g = s0 + idx;
    x_s[idx] = (g >= 0 && g < L) ? (float)xr[g] : 0.0f;
  }
  __syncthreads();
  scalar_t* yr = y + ((long)n * Co + co) * Lo;
  const float b0 = HAS_BIAS ? (float)bias[co] : 0.0f;
#pragma unroll
  for (int t = 0; t < kDwTile; ++t) {
    const long lo = lo0 + t * kBlock + threadIdx.x;
    if (lo >= Lo) break;
    float acc = b0;
    const int base = (t * kBlock + threadIdx.x) * stride;
    for (int k = 0; k < K; ++k) {
      acc += w_lds[k] * x_s[base + k * dil];
    }
    yr[lo] = (scalar_t)acc;
  }
}

template <typename scalar_t, int STRIDE>
__global__ void dwconv_dx_kernel(const scalar_t* __restrict__ dy,
                                 const scalar_t* __restrict__ w,
                                 scalar_t* __restrict__ dx,
                                 int N, int Ci, int Co, long L, long Lo,
                                 int K, int stride, int padl, int dil) {
  extern __shared__ float w_lds[];  // [K] floats, then the dy window
  float* dy_s = w_lds + K;
  const int n = blockIdx.y;
  const int ci = blockIdx.z;
  const long li0 = (long)blockIdx.x * (kBlock * kDwTile);
  const int s = (STRIDE > 0) ? STRIDE : stride;
  // lo range touched by li in [li0, li0+kBlock*kDwTile): stage with one
  // element of slack on each side
  const long lob = (li0 + padl - (long)(K - 1) * dil) / (long)s - 1;
  const int ext = (kBlock * kDwTile) / s + ((K - 1) * dil) / s + 4;
  const scalar_t* dyr = dy + ((long)n * Co + ci) * Lo;
  for (int idx = threadIdx.x; idx < K; idx += kBlock) {
    w_lds[idx] = (float)w[(long)ci * K + idx];
  }
  for (int idx = threadIdx.x; idx < ext; idx += kBlock) {
    const long g = lob + idx;
    dy_s[idx] = (g >= 0 && g < Lo) ? (float)dyr[g] : 0.0f;
  }
  __syncthreads();
  scalar_t* dxr = dx + ((long)n * Ci + ci) * L;
#pragma unroll
  for (int t = 0; t < kDwTile; ++t) {
    const long li = li0 + t * kBlock + threadIdx.x;
    if (li >= L) break;
    float acc = 0.0f;
    if (STRIDE != 1 && dil == 1) {
      // phase decomposition: only k == (li+padl) mod s hits a valid lo
      for (int k = (int)((li + padl) % s); k < K; k += s) {
        const long lo = (li + padl - k) / s;
        if (lo >= 0 && lo < Lo) acc += w_lds[k] * dy_s[lo - lob];
      }
    } else {
      for (int k = 0; k < K; ++k) {
        const long num = li + padl - (long)k * dil;
        if (num < 0) continue;
        if (STRIDE != 1 && (num % s)) continue;
        const long lo = (STRIDE == 1) ? num : num / s;
        if (lo < Lo) acc += w_lds[k] * dy_s[lo - lob];
      }
    }
    dxr[li] = (scalar_t)acc;
  }
}

// ---------------- dense / grouped conv-GEMM ----------------
// block: (lo-chunk, n, group-co-chunk). Weights for the 32-co chunk live
// in LDS ([jo][cig_chunk*K], padded); each staged x value feeds 32 FMAs.
template <typename scalar_t, bool HAS_BIAS>
__global__ void convgemm_fwd_kernel(const scalar_t* __restrict__ x,
                                    const scalar_t* __restrict__ w,
                                    const scalar_t* __restrict__ bias,
                                    scalar_t* __restrict__ y,
                                    int N, int Ci, int Co, long L, long Lo,
                                    int K, int stride, int padl, int dil,
                                    int G, int cig_chunk) {
  extern __shared__ float w_lds[];  // [kJT][cig_chunk*K]
  const int n = blockIdx.y;
  const int Cog = Co / G;
  const int Cig = Ci / G;
  const int chunks_per_g = (Cog + kJT - 1) / kJT;
  const int g = blockIdx.z / chunks_per_g;
  const int j0 = (blockIdx.z - g * chunks_per_g) * kJT;
  const int co0 = g * Cog + j0;
  const int jn = min(kJT, Cog - j0);
  const long lo = (long)blockIdx.x * kBlock + threadIdx.x;
  const int wrow = cig_chunk * K;

  float acc[kJT];
#pragma unroll
  for (int j = 0; j < kJT; ++j) acc[j] = 0.0f;

  const long li0 = lo * stride - padl;
  for (int c0 = 0; c0 < Cig; c0 += cig_chunk) {
    const int cn = min(cig_chunk, Cig - c0);
    __syncthreads();
    for (int idx = threadIdx.x; idx < jn * cn * K; idx += kBlock) {
      const int jo = idx / (cn * K);
      const int r = idx - jo * cn * K;  // cig*K + k within chunk
      w_lds[jo * wrow + r] = (float)w[((long)(co0 + jo) * Cig + c0) * K + r];
    }
    __syncthreads();
    if (lo < Lo) {
      const scalar_t* xb = x + ((long)n * Ci + (long)g * Cig + c0) * L;
      for (int c = 0; c < cn; ++c) {
        const scalar_t* xr = xb + (long)c * L;
        for (int k = 0; k < K; ++k) {
          const long li = li0 + (long)k * dil;
          if (li < 0 || li >= L) continue;
          const float xv = (float)xr[li];
          const float* wp = w_lds + c * K + k;
#pragma unroll
          for (int j = 0; j < kJT; ++j) {
            acc[j] += wp[j * wrow] * xv;
          }
        }
      }
    }
  }

  if (lo < Lo) {
    scalar_t* yp = y + ((long)n * Co + co0) * Lo + lo;
    for (int j = 0; j < jn; ++j) {
      float v = acc[j];
      if (HAS_BIAS) v += (float)bias[co0 + j];
      yp[(long)j * Lo] = (scalar_t)v;
    }
  }
}

// dx chunked the same way: 32 input channels per block share dy reads.
template <typename scalar_t>
__global__ void convgemm_dx_kernel(const scalar_t* __restrict__ dy,
                                   const scalar_t* __restrict__ w,
                                   scalar_t* __restrict__ dx,
                                   int N, int Ci, int Co, long L, long Lo,
                                   int K, int stride, int padl, int dil,
                                   int G, int cog_chunk) {
  extern __shared__ float w_lds[];  // [kJT(ci)][cog_chunk*K]
  const int n = blockIdx.y;
  const int Cog = Co / G;
  const int Cig = Ci / G;
  const int chunks_per_g = (Cig + kJT - 1) / kJT;
  const int g = blockIdx.z / chunks_per_g;
  const int i0 = (blockIdx.z - g * chunks_per_g) * kJT;
  const int ci0 = g * Cig + i0;
  const int in_ = min(kJT, Cig - i0);
  const long li = (long)blockIdx.x * kBlock + threadIdx.x;
  const int wrow = cog_chunk * K;

  float acc[kJT];
#pragma unroll
  for (int j = 0; j < kJT; ++j) acc[j] = 0.0f;

  for (int j0_ = 0; j0_ < Cog; j0_ += cog_chunk) {
    const int jn = min(cog_chunk, Cog - j0_);
    __syncthreads();
    for (int idx = threadIdx.x; idx < in_ * jn * K; idx += kBlock) {
      const int ii = idx / (jn * K);
      const int r = idx - ii * jn * K;
      const int jo = r / K;
      const int k = r - jo * K;
      w_lds[ii * wrow + r] =
          (float)w[(((long)(g * Cog + j0_ + jo)) * Cig + i0 + ii) * K + k];
    }
    __syncthreads();
    if (li < L) {
      // phase decomposition: with dil==1 only taps k == (li+padl) mod s
      // divide evenly — iterate those directly (no per-tap modulo)
      const int kstart = (dil == 1 && stride > 1)
                             ? (int)((li + padl) % stride) : 0;
      const int kstep = (dil == 1 && stride > 1) ? stride : 1;
      for (int jo = 0; jo < jn; ++jo) {
        const scalar_t* dyr = dy + ((long)n * Co + g * Cog + j0_ + jo) * Lo;
        for (int k = kstart; k < K; k += kstep) {
          const long num = li + padl - (long)k * dil;
          if (num < 0) continue;
          if (kstep == 1 && stride > 1 && (num % stride)) continue;
          const long lo = (stride > 1) ? num / stride : num;
          if (lo >= Lo) continue;
          const float dyv = (float)dyr[lo];
          const float* wp = w_lds + jo * K + k;
#pragma unroll
          for (int j = 0; j < kJT; ++j) {
            acc[j] += wp[j * wrow] * dyv;
          }
        }
      }
    }
  }

  if (li < L) {
    scalar_t* dxp = dx + ((long)n * Ci + ci0) * L + li;
    for (int j = 0; j < in_; ++j) {
      dxp[(long)j * L] = (scalar_t)acc[j];
    }
  }
}

// ---------------- weight gradient ----------------
// One block per (co, cig-pair, n-split): dy chunk staged in LDS once and
// reused for every (cig, k); KT is the compile-time tap bound so the
// accumulators stay in registers.
constexpr int kMaxK = 24;

template <typename scalar_t, bool HAS_BIAS, int KT, int kCigT>
__global__ void conv1d_dw_kernel(const scalar_t* __restrict__ dy,
                                 const scalar_t* __restrict__ x,
                                 float* __restrict__ dw,
                                 float* __restrict__ db,
                                 int N, int Ci, int Co, long L, long Lo,
                                 int K, int stride, int padl, int dil,
                                 int G, int nsplit) {
  __shared__ float dy_s[kBlock];
  __shared__ float red[kBlock / sa::kWave];

  const int co = blockIdx.x;
  const int cig0 = blockIdx.y * kCigT;
  const int Cig = Ci / G;
  const int g = co / (Co / G);
  const int cig_n = min(kCigT, Cig - cig0);

  const long nchunk = ((long)N + nsplit - 1) / nsplit;
  const long n0 = (long)blockIdx.z * nchunk;
  const long n1 = min((long)N, n0 + nchunk);

  float acc[kCigT][KT];
#pragma unroll
  for (int c = 0; c < kCigT; ++c)
#pragma unroll
    for (int k = 0; k < KT; ++k) acc[c][k] = 0.0f;
  float bacc = 0.0f;

  for (long n = n0; n < n1; ++n) {
    const scalar_t* dyr = dy + ((long)n * Co + co) * Lo;
    const scalar_t* xr0 = x + ((long)n * Ci + g * Cig + cig0) * L;
    for (long lo0 = 0; lo0 < Lo; lo0 += kBlock) {
      const long lo = lo0 + threadIdx.x;
      __syncthreads();
      dy_s[threadIdx.x] = (lo < Lo) ? (float)dyr[lo] : 0.0f;
      __syncthreads();
      if (lo < Lo) {
        const float dyv = dy_s[threadIdx.x];
        if (HAS_BIAS && blockIdx.y == 0) bacc += dyv;
        const long li0 = lo * stride - padl;
#pragma unroll
        for (int c = 0; c < kCigT; ++c) {
          if (c >= cig_n) break;
          const scalar_t* xr = xr0 + (long)c * L;
#pragma unroll
          for (int k = 0; k < KT; ++k) {
            if (k >= K) break;
            const long li = li0 + (long)k * dil;
            if (li >= 0 && li < L) {
              acc[c][k] += dyv * (float)xr[li];
            }
          }
        }
      }
    }
  }

#pragma unroll
  for (int c = 0; c < kCigT; ++c) {
#pragma unroll
    for (int k = 0; k < KT; ++k) {
      if (c < cig_n && k < K) {
        __syncthreads();
        const float v = sa::block_reduce_sum(acc[c][k], red);
        if (threadIdx.x == 0) {
          atomicAdd(&dw[((long)co * Cig + cig0 + c) * K + k], v);
        }
      }
    }
  }
  if (HAS_BIAS && blockIdx.y == 0) {
    __syncthreads();
    const float v = sa::block_reduce_sum(bacc, red);
    if (threadIdx.x == 0) atomicAdd(&db[co], v);
  }
}

int pick_cig_chunk(int Cig, int K) {
  int chunk = Cig;
  while (kJT * chunk * K > kMaxWLds) chunk = (chunk + 1) / 2;
  return std::max(1, chunk);
}

}  // namespace

at::Tensor conv1d_fwd(const at::Tensor& x, const at::Tensor& w,
                      const c10::optional<at::Tensor>& bias, long stride,
                      long padl, long padr, long groups, long dilation) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous());
  const int N = x.size(0), Ci = x.size(1);
  const long L = x.size(2);
  const int Co = w.size(0), K = w.size(2);
  TORCH_CHECK((long)w.size(1) * groups == Ci, "group/channel mismatch");
  const long Lp = L + padl + padr;
  const long Lo = (Lp - ((long)K - 1) * dilation - 1) / stride + 1;
  TORCH_CHECK(Lo > 0, "empty conv output");
  auto y = at::empty({N, Co, Lo}, x.options());

  const bool has_bias = bias.has_value() && bias->defined();
  at::Tensor bct;
  if (has_bias) bct = bias->to(x.scalar_type()).contiguous();

  const int Cig = Ci / groups;
  const int Cog = Co / groups;
  auto stream = at::hip::getCurrentHIPStream();

  // stride-1 convs (dense AND grouped) go to the matrix cores: the
  // tap-gather kernel stages each channel row once; the im2col kernel
  // remains as the fp32 fallback
  if (stride == 1
      && conv_smallc_mfma(x, w, bias, y, padl, dilation, groups,
                          /*is_dx=*/false)) {
    return y;
  }
  if (stride == 1
      && conv_tap_mfma(x, w, bias, y, padl, dilation, groups,
                       /*is_dx=*/false)) {
    return y;
  }
  if (conv_tap_s_mfma(x, w, bias, y, stride, padl, dilation, groups,
                      /*is_dx=*/false)) {
    return y;
  }
  if (groups == 1 && stride == 1
      && conv_mfma(x, w, bias, y, padl, dilation, /*is_dx=*/false)) {
    return y;
  }

  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "conv1d_fwd", [&] {
        const scalar_t* bp = has_bias ? bct.data_ptr<scalar_t>() : nullptr;
        if (Cog == 1 && Cig == 1) {  // true depthwise (groups == Ci == Co)
          dim3 grid(sa::ceil_div(Lo, (long)kBlock * kDwTile), N, Co);
          const size_t lds = sizeof(float) *
              (K + kBlock * kDwTile * stride + (K - 1) * dilation + 1);
          TORCH_CHECK(lds <= 64 * 1024, "depthwise LDS window too large");
          if (has_bias) {
            hipLaunchKernelGGL((dwconv_fwd_kernel<scalar_t, true>), grid,
                               dim3(kBlock), lds, stream.stream(),
                               x.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                               bp, y.data_ptr<scalar_t>(), N, Ci, Co, L, Lo,
                               K, (int)stride, (int)padl, (int)dilation);
          } else {
            hipLaunchKernelGGL((dwconv_fwd_kernel<scalar_t, false>), grid,
                               dim3(kBlock), lds, stream.stream(),
                               x.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                               bp, y.data_ptr<scalar_t>(), N, Ci, Co, L, Lo,
                               K, (int)stride, (int)padl, (int)dilation);
          }
        } else {
          const int cig_chunk = pick_cig_chunk(Cig, K);
          const int chunks_per_g = sa::ceil_div(Cog, kJT);
          dim3 grid(sa::ceil_div(Lo, kBlock), N,
                    (int)groups * chunks_per_g);
          const size_t lds = sizeof(float) * kJT * cig_chunk * K;
          if (has_bias) {
            hipLaunchKernelGGL((convgemm_fwd_kernel<scalar_t, true>), grid,
                               dim3(kBlock), lds, stream.stream(),
                               x.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                               bp, y.data_ptr<scalar_t>(), N, Ci, Co, L, Lo,
                               K, (int)stride, (int)padl, (int)dilation,
                               (int)groups, cig_chunk);
          } else {
            hipLaunchKernelGGL((convgemm_fwd_kernel<scalar_t, false>), grid,
                               dim3(kBlock), lds, stream.stream(),
                               x.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                               bp, y.data_ptr<scalar_t>(), N, Ci, Co, L, Lo,
                               K, (int)stride, (int)padl, (int)dilation,
                               (int)groups, cig_chunk);
          }
        }
      });
  return y;
}

// forward + per-out-channel (sum, sumsq) BN partials (fusion step 1);
// falls back to conv1d_fwd + a bn_sums pass outside the MFMA envelope.
std::vector<at::Tensor> conv1d_fwd_stats(
    const at::Tensor& x, const at::Tensor& w,
    const c10::optional<at::Tensor>& bias, long stride, long padl,
    long padr, long groups, long dilation) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous());
  const int N = x.size(0);
  const long L = x.size(2);
  const int Co = w.size(0), K = w.size(2);
  const long Lp = L + padl + padr;
  const long Lo = (Lp - ((long)K - 1) * dilation - 1) / stride + 1;
  TORCH_CHECK(Lo > 0, "empty conv output");
  auto y = at::empty({N, Co, Lo}, x.options());
  at::Tensor part;
  if (stride == 1
      && conv_smallc_mfma(x, w, bias, y, padl, dilation, groups,
                          /*is_dx=*/false, &part)) {
    return {y, part};
  }
  if (stride == 1
      && conv_tap_mfma(x, w, bias, y, padl, dilation, groups,
                       /*is_dx=*/false, &part)) {
    return {y, part};
  }
  if (conv_tap_s_mfma(x, w, bias, y, stride, padl, dilation, groups,
                      /*is_dx=*/false, &part)) {
    return {y, part};
  }
  if (groups == 1 && stride == 1
      && conv_mfma(x, w, bias, y, padl, dilation, /*is_dx=*/false, &part)) {
    return {y, part};
  }
  y = conv1d_fwd(x, w, bias, stride, padl, padr, groups, dilation);
  part = bn_sums_only(y).view({1, Co, 2});
  return {y, part};
}

// Gather pass shared by the conv input-gradient and the transposed-conv
// forward: out[n][ci][li] = sum_{co in group, k} in[n][co][(li + padl -
// k*d)/stride] * w[co][cig][k] (only where divisible/in range).
void conv1d_dx_into(const at::Tensor& dy, const at::Tensor& w,
                    at::Tensor& dx, long stride, long padl, long groups,
                    long dilation) {
  const int N = dx.size(0), Ci = dx.size(1);
  const long L = dx.size(2);
  const int Co = w.size(0), K = w.size(2);
  const long Lo = dy.size(2);
  const int Cig = Ci / groups;
  const int Cog = Co / groups;
  auto stream = at::hip::getCurrentHIPStream();

  if (stride == 1
      && conv_smallc_mfma(dy, w, c10::nullopt, dx, padl, dilation, groups,
                          /*is_dx=*/true)) {
    return;
  }
  if (stride == 1
      && conv_tap_mfma(dy, w, c10::nullopt, dx, padl, dilation, groups,
                       /*is_dx=*/true)) {
    return;
  }
  if (conv_tap_s_mfma(dy, w, c10::nullopt, dx, stride, padl, dilation,
                      groups, /*is_dx=*/true)) {
    return;
  }
  if (groups == 1 && stride == 1
      && conv_mfma(dy, w, c10::nullopt, dx, padl, dilation, /*is_dx=*/true)) {
    return;
  }
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, dx.scalar_type(),
      "conv1d_dx", [&] {
        if (Cog == 1 && Cig == 1) {
          dim3 grid(sa::ceil_div(L, (long)kBlock * kDwTile), N, Ci);
          const size_t lds = sizeof(float) *
              (K + (kBlock * kDwTile) / stride
               + ((K - 1) * dilation) / stride + 4);
          TORCH_CHECK(lds <= 64 * 1024, "depthwise LDS window too large");
          auto launch_dw = [&](auto st) {
            hipLaunchKernelGGL((dwconv_dx_kernel<scalar_t,
                                                 decltype(st)::value>),
                               grid, dim3(kBlock), lds, stream.stream(),
                               dy.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                               dx.data_ptr<scalar_t>(), N, Ci, Co, L, Lo, K,
                               (int)stride, (int)padl, (int)dilation);
          };
          if (stride == 1) launch_dw(std::integral_constant<int, 1>{});
          else if (stride == 2) launch_dw(std::integral_constant<int, 2>{});
          else launch_dw(std::integral_constant<int, 0>{});
        } else {
          const int cog_chunk = pick_cig_chunk(Cog, K);
          const int chunks_per_g = sa::ceil_div(Cig, kJT);
          dim3 grid(sa::ceil_div(L, kBlock), N, (int)groups * chunks_per_g);
          const size_t lds = sizeof(float) * kJT * cog_chunk * K;
          hipLaunchKernelGGL((convgemm_dx_kernel<scalar_t>), grid,
                             dim3(kBlock), lds, stream.stream(),
                             dy.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                             dx.data_ptr<scalar_t>(), N, Ci, Co, L, Lo, K,
                             (int)stride, (int)padl, (int)dilation,
                             (int)groups, cog_chunk);
        }
      });
}

std::vector<at::Tensor> conv1d_dw_db(const at::Tensor& dy,
                                     const at::Tensor& x,
                                     const at::Tensor& w, long stride,
                                     long padl, long padr, long groups,
                                     long dilation, bool has_bias) {
  const int N = x.size(0), Ci = x.size(1);
  const long L = x.size(2);
  const int Co = w.size(0), K = w.size(2);
  const long Lo = dy.size(2);
  const int Cig = Ci / groups;
  const int Cog = Co / groups;
  auto stream = at::hip::getCurrentHIPStream();

  // stride-1 convs (dense AND grouped): dw is K shifted plain GEMMs — run
  // them on the matrix cores via rocBLAS strided-batched bmm over zero-copy
  // strided views. For groups > 1 the (N, G*Cog, l) slice is viewed as an
  // (N*G)-batch of (Cog, l) panels (the N and G strides compose exactly),
  // so one batched GEMM covers every group.
  // grouped convs keep the direct kernel: at Cog=Cig=8 the per-tap
  // batched GEMM is launch/overhead-bound and measured slower.
  // small-channel dense convs: one MFMA kernel + one batch sum replaces
  // the K-tap bmm slab + sum_mid chain (measured ~4-9 ms/step of GEMM +
  // reduce on eqt/ditingmotion; profiles/step_profile_r02.md)
  if (auto dws = conv_dw_smallc(dy, x, K, padl, stride, dilation, groups,
                                w.scalar_type());
      dws.defined()) {
    at::Tensor db;
    if (has_bias) db = channel_sum_to(dy, w.scalar_type());
    return {dws, db};
  }

  if (groups == 1 && stride == 1) {
    // one bmm per tap into a (K, N, Co, Ci) slab, one middle-axis sum over
    // N for all taps, one strided scatter into the (Co, Ci, K) layout
    auto slab = at::empty({(long)K, (long)N, (long)Co, (long)Cig},
                          x.options());
    for (int k = 0; k < K; ++k) {
      const long off = (long)k * dilation - padl;
      const long lo0 = std::max<long>(0, -off);
      const long lo1 = std::min<long>(Lo, L - off);
      if (lo1 <= lo0) {
        slab.select(0, k).zero_();  // fully clipped tap
        continue;
      }
      const long l = lo1 - lo0;
      auto dyv = at::as_strided(dy, {(long)N, (long)Co, l},
                                {(long)Co * Lo, Lo, 1},
                                dy.storage_offset() + lo0);
      auto xv = at::as_strided(x, {(long)N, (long)Cig, l},
                               {(long)Cig * L, L, 1},
                               x.storage_offset() + lo0 + off);
      auto out_k = slab.select(0, k);
      at::bmm_out(out_k, dyv, xv.transpose(1, 2));
    }
    auto dw = sum_mid_to(slab.view({(long)K, (long)N, (long)Co * Cig}),
                         w.scalar_type())
                  .view({(long)K, (long)Co, (long)Cig})
                  .permute({1, 2, 0})
                  .contiguous();
    at::Tensor db;
    if (has_bias) {
      db = channel_sum_to(dy, w.scalar_type());
    }
    return {dw, db};
  }

  // strided dense convs (phasenet encoder / transposed-conv grad): the
  // direct accumulation kernel below is ~100x off roofline here (measured
  // 1.25 ms/call at Co=8, profiles/step_profile_r02.md). One padded copy
  // of x makes every tap an arithmetic-sequence view, so a zero-copy
  // (N, Cig, K, Lo) as_strided with strides (Ci*Lp, Lp, dil, s)
  // materialized once (= im2col) turns dw into ONE batched GEMM over the
  // matrix cores plus one batch-axis sum.
  if (groups == 1 && stride > 1
      && (long)N * Cig * K * Lo <= (long)256 * 1024 * 1024) {
    auto xp = (padl > 0 || padr > 0)
        ? at::constant_pad_nd(x, {padl, padr})
        : x;
    const long Lp = xp.size(2);
    auto xs = at::as_strided(xp, {(long)N, (long)Cig, (long)K, Lo},
                             {(long)Ci * Lp, Lp, dilation, stride});
    auto xc = xs.reshape({(long)N, (long)Cig * K, Lo});  // one im2col copy
    auto slab = at::bmm(dy, xc.transpose(1, 2));         // (N, Co, Cig*K)
    auto dw = sum_batch_to(slab.view({(long)N, (long)Co * Cig * K}),
                           w.scalar_type())
                  .view({(long)Co, (long)Cig, (long)K});
    at::Tensor db;
    if (has_bias) db = channel_sum_to(dy, w.scalar_type());
    return {dw, db};
  }

  // grouped/depthwise stride-1 convs: matrix-core weight gradient
  if (auto dwm = dw_mfma_try(dy, x, stride, padl, groups, dilation, K)) {
    at::Tensor db;
    if (has_bias) db = channel_sum_to(dy, w.scalar_type());
    return {dwm->to(w.scalar_type()), db};
  }

  auto dw32 = at::zeros_like(w, w.options().dtype(at::kFloat));
  at::Tensor db32;
  if (has_bias) db32 = at::zeros({Co}, w.options().dtype(at::kFloat));
  {
    TORCH_CHECK(K <= kMaxK, "conv1d_dw: kernel taps > ", kMaxK);
    const int cig_t = (K <= 4) ? 8 : (K <= 8) ? 4 : 2;
    const int nsplit = std::max(1, std::min<int>(
        (int)N, 4096 / (Co * sa::ceil_div(Cig, cig_t))));
    dim3 grid(Co, sa::ceil_div(Cig, cig_t), nsplit);
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
        "conv1d_dw", [&] {
          auto launch_kt = [&](auto kt, auto ct, auto hb) {
            hipLaunchKernelGGL(
                (conv1d_dw_kernel<scalar_t, decltype(hb)::value,
                                  decltype(kt)::value, decltype(ct)::value>),
                grid, dim3(kBlock), 0, stream.stream(),
                dy.data_ptr<scalar_t>(), x.data_ptr<scalar_t>(),
                dw32.data_ptr<float>(),
                decltype(hb)::value ? db32.data_ptr<float>() : nullptr,
                N, Ci, Co, L, Lo, K, (int)stride, (int)padl,
                (int)dilation, (int)groups, nsplit);
          };
          auto launch_hb = [&](auto kt, auto ct) {
            if (has_bias) launch_kt(kt, ct, std::true_type{});
            else launch_kt(kt, ct, std::false_type{});
          };
          // accumulator budget ~32 fp32: wider cig tiles at small K
          if (K <= 4) launch_hb(std::integral_constant<int, 4>{},
                                std::integral_constant<int, 8>{});
          else if (K <= 8) launch_hb(std::integral_constant<int, 8>{},
                                     std::integral_constant<int, 4>{});
          else if (K <= 12) launch_hb(std::integral_constant<int, 12>{},
                                      std::integral_constant<int, 2>{});
          else launch_hb(std::integral_constant<int, kMaxK>{},
                         std::integral_constant<int, 2>{});
        });
  }
  auto dw = dw32.to(w.scalar_type());
  at::Tensor db;
  if (has_bias) db = db32.to(w.scalar_type());
  return {dw, db};
}

std::vector<at::Tensor> conv1d_bwd(const at::Tensor& dy, const at::Tensor& x,
                                   const at::Tensor& w, long stride,
                                   long padl, long padr, long groups,
                                   long dilation, bool has_bias) {
  auto dx = at::empty_like(x);
  conv1d_dx_into(dy, w, dx, stride, padl, groups, dilation);
  auto dwdb = conv1d_dw_db(dy, x, w, stride, padl, padr, groups, dilation,
                           has_bias);
  return {dx, dwdb[0], dwdb[1]};
}

// ---------------------------------------------------------------------------
// ConvTranspose1d (K5) — PhaseNet's up path (reference phasenet.py:118-127).
// Forward is the conv input-gradient gather with roles swapped; the input
// gradient is a strided conv forward of dY with w viewed (Ci, Co, K) — the
// exact layout ConvTranspose1d stores; the weight gradient is the conv
// weight gradient with x and dY swapped.
// ---------------------------------------------------------------------------

at::Tensor conv_transpose1d_fwd(const at::Tensor& x, const at::Tensor& w,
                                const c10::optional<at::Tensor>& bias,
                                long stride) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && w.is_contiguous());
  const int N = x.size(0);
  const int Ci = x.size(1);   // convT input channels
  const long Li = x.size(2);
  const int Co = w.size(1);   // w: (Ci, Co, K)
  const int K = w.size(2);
  TORCH_CHECK(w.size(0) == Ci, "conv_transpose weight/input mismatch");
  const long Lo = (Li - 1) * stride + K;
  auto y = at::empty({(long)N, (long)Co, Lo}, x.options());
  // view w (Ci, Co, K) as a conv weight (Co_conv=Ci, Cig=Co, K)
  conv1d_dx_into(/*dy=*/x, /*w=*/w, /*dx=*/y, stride, /*padl=*/0,
                 /*groups=*/1, /*dilation=*/1);
  if (bias.has_value() && bias->defined()) {
    y.add_(bias->to(y.scalar_type()).view({1, (long)Co, 1}));
  }
  return y;
}

std::vector<at::Tensor> conv_transpose1d_bwd(const at::Tensor& dy,
                                             const at::Tensor& x,
                                             const at::Tensor& w,
                                             long stride, bool has_bias) {
  // dX = strided conv forward of dY with weight (Ci, Co, K) as-is
  auto dx = conv1d_fwd(dy, w, c10::nullopt, stride, /*padl=*/0,
                       /*padr=*/0, /*groups=*/1, /*dilation=*/1);
  // dW[ci][co][k]: conv weight gradient with (dy:=x, x:=dY)
  auto dws = conv1d_dw_db(/*dy=*/x, /*x=*/dy, /*w=*/w, stride, /*padl=*/0,
                          /*padr=*/0, /*groups=*/1, /*dilation=*/1,
                          /*has_bias=*/false);
  auto dw = dws[0];
  at::Tensor db;
  if (has_bias) db = channel_sum_to(dy, w.scalar_type());
  return {dx, dw, db};
}
