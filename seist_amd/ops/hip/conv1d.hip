// Direct Conv1d (dense / grouped / depthwise / dilated-causal) —
// K2/K3/K4/K6 of SURVEY.md §2.4. Covers the reference's depthwise stem
// convs (models/seist.py:134-141), grouped convs (:215-222), dense head /
// PhaseNet / EQT convs, and dist-PT's dilated causal convs
// (models/distpt_network.py:17-87).
//
// x (N, Ci, L), w (Co, Cig=Ci/G, K), y (N, Co, Lo)
// y[n][co][lo] = sum_{cig,k} w[co][cig][k] * x[n][g*Cig+cig][lo*s - padl + k*d]
//
// Weights are staged in LDS (Cig*K <= a few KB for every model in the
// zoo); each block computes 256 consecutive lo of one (n, co) row, so all
// global traffic is coalesced and each x row is read once per k-tap from
// L1/L2.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kMaxWLds = 4096;  // floats of LDS weight stage per block

template <typename scalar_t, bool HAS_BIAS>
__global__ void conv1d_fwd_kernel(const scalar_t* __restrict__ x,
                                  const scalar_t* __restrict__ w,
                                  const float* __restrict__ bias,
                                  scalar_t* __restrict__ y,
                                  int N, int Ci, int Co, long L, long Lo,
                                  int K, int stride, int padl, int dil,
                                  int G) {
  extern __shared__ float w_lds[];  // [Cig][K] for this co

  const int n = blockIdx.y;
  const int co = blockIdx.z;
  const int Cig = Ci / G;
  const int g = co / (Co / G);
  const long lo = (long)blockIdx.x * kBlock + threadIdx.x;

  for (int idx = threadIdx.x; idx < Cig * K; idx += kBlock) {
    w_lds[idx] = (float)w[(long)co * Cig * K + idx];
  }
  __syncthreads();

  if (lo >= Lo) return;

  const long li0 = lo * stride - padl;
  float acc = HAS_BIAS ? bias[co] : 0.0f;
  const scalar_t* xb = x + ((long)n * Ci + (long)g * Cig) * L;
  for (int cig = 0; cig < Cig; ++cig) {
    const scalar_t* xr = xb + (long)cig * L;
    const float* wr = w_lds + cig * K;
    for (int k = 0; k < K; ++k) {
      const long li = li0 + (long)k * dil;
      if (li >= 0 && li < L) acc += wr[k] * (float)xr[li];
    }
  }
  y[((long)n * Co + co) * Lo + lo] = (scalar_t)acc;
}

// dx[n][ci][li] = sum_{co in group, k} dy[n][co][lo] * w[co][cig][k]
//   where lo = (li + padl - k*d) / s  (when divisible and in range)
template <typename scalar_t>
__global__ void conv1d_dx_kernel(const scalar_t* __restrict__ dy,
                                 const scalar_t* __restrict__ w,
                                 scalar_t* __restrict__ dx,
                                 int N, int Ci, int Co, long L, long Lo,
                                 int K, int stride, int padl, int dil,
                                 int G) {
  extern __shared__ float w_lds[];  // [Cog][K] for this ci's group

  const int n = blockIdx.y;
  const int ci = blockIdx.z;
  const int Cig = Ci / G;
  const int Cog = Co / G;
  const int g = ci / Cig;
  const int cig = ci - g * Cig;
  const long li = (long)blockIdx.x * kBlock + threadIdx.x;

  // stage w[g*Cog + j][cig][k] for j in [0, Cog)
  for (int idx = threadIdx.x; idx < Cog * K; idx += kBlock) {
    const int j = idx / K;
    const int k = idx - j * K;
    w_lds[idx] = (float)w[(((long)(g * Cog + j)) * Cig + cig) * K + k];
  }
  __syncthreads();

  if (li >= L) return;

  float acc = 0.0f;
  for (int j = 0; j < Cog; ++j) {
    const scalar_t* dyr = dy + ((long)n * Co + g * Cog + j) * Lo;
    const float* wr = w_lds + j * K;
    for (int k = 0; k < K; ++k) {
      const long num = li + padl - (long)k * dil;
      if (num < 0) continue;
      if (num % stride) continue;
      const long lo = num / stride;
      if (lo < Lo) acc += wr[k] * (float)dyr[lo];
    }
  }
  dx[((long)n * Ci + ci) * L + li] = (scalar_t)acc;
}

// dw[co][cig][k] = sum_{n,lo} dy[n][co][lo] * x[n][g*Cig+cig][lo*s-padl+k*d]
//
// One block per (co, cig-pair, n-split). The dy row chunk is staged once
// in LDS and reused for every (cig, k); each thread accumulates all
// CIG_T*K tap products for its own lo positions, then the block reduces.
// This reads x only Co times and dy only ceil(Cig/CIG_T) times — the
// previous per-(co,cig,k)-block design read x Co*K times and was the
// second-largest kernel cost of the training step.
constexpr int kCigT = 2;    // cig channels per block
constexpr int kMaxK = 24;   // max kernel taps supported (zoo max is 19)

// KT is the compile-time tap-count bound so the accumulator array stays in
// registers (runtime-indexed register arrays spill to scratch on gfx950).
template <typename scalar_t, bool HAS_BIAS, int KT>
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

  // block-reduce each accumulator and emit
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

  at::Tensor b32;
  const bool has_bias = bias.has_value() && bias->defined();
  if (has_bias) b32 = bias->to(at::kFloat).contiguous();

  const int Cig = Ci / groups;
  TORCH_CHECK(Cig * K <= kMaxWLds, "weight tile too large for LDS stage");
  const size_t lds = sizeof(float) * Cig * K;
  dim3 grid(sa::ceil_div(Lo, kBlock), N, Co);
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "conv1d_fwd", [&] {
        if (has_bias) {
          hipLaunchKernelGGL((conv1d_fwd_kernel<scalar_t, true>), grid,
                             dim3(kBlock), lds, stream.stream(),
                             x.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                             b32.data_ptr<float>(), y.data_ptr<scalar_t>(),
                             N, Ci, Co, L, Lo, K, (int)stride, (int)padl,
                             (int)dilation, (int)groups);
        } else {
          hipLaunchKernelGGL((conv1d_fwd_kernel<scalar_t, false>), grid,
                             dim3(kBlock), lds, stream.stream(),
                             x.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                             nullptr, y.data_ptr<scalar_t>(),
                             N, Ci, Co, L, Lo, K, (int)stride, (int)padl,
                             (int)dilation, (int)groups);
        }
      });
  return y;
}

std::vector<at::Tensor> conv1d_bwd(const at::Tensor& dy, const at::Tensor& x,
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

  auto dx = at::empty_like(x);
  {
    TORCH_CHECK(Cog * K <= kMaxWLds, "weight tile too large for LDS stage");
    const size_t lds = sizeof(float) * Cog * K;
    dim3 grid(sa::ceil_div(L, kBlock), N, Ci);
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
        "conv1d_dx", [&] {
          hipLaunchKernelGGL((conv1d_dx_kernel<scalar_t>), grid,
                             dim3(kBlock), lds, stream.stream(),
                             dy.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                             dx.data_ptr<scalar_t>(),
                             N, Ci, Co, L, Lo, K, (int)stride, (int)padl,
                             (int)dilation, (int)groups);
        });
  }

  auto dw32 = at::zeros_like(w, w.options().dtype(at::kFloat));
  at::Tensor db32;
  if (has_bias) db32 = at::zeros({Co}, w.options().dtype(at::kFloat));
  {
    TORCH_CHECK(K <= kMaxK, "conv1d_dw: kernel taps > ", kMaxK);
    const int nsplit = std::max(1, std::min<int>(
        (int)N, 4096 / (Co * sa::ceil_div(Cig, kCigT))));
    dim3 grid(Co, sa::ceil_div(Cig, kCigT), nsplit);
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
        "conv1d_dw", [&] {
          auto launch_kt = [&](auto kt, auto hb) {
            hipLaunchKernelGGL(
                (conv1d_dw_kernel<scalar_t, decltype(hb)::value,
                                  decltype(kt)::value>),
                grid, dim3(kBlock), 0, stream.stream(),
                dy.data_ptr<scalar_t>(), x.data_ptr<scalar_t>(),
                dw32.data_ptr<float>(),
                decltype(hb)::value ? db32.data_ptr<float>() : nullptr,
                N, Ci, Co, L, Lo, K, (int)stride, (int)padl,
                (int)dilation, (int)groups, nsplit);
          };
          auto launch_hb = [&](auto kt) {
            if (has_bias) launch_kt(kt, std::true_type{});
            else launch_kt(kt, std::false_type{});
          };
          if (K <= 4) launch_hb(std::integral_constant<int, 4>{});
          else if (K <= 8) launch_hb(std::integral_constant<int, 8>{});
          else if (K <= 12) launch_hb(std::integral_constant<int, 12>{});
          else launch_hb(std::integral_constant<int, kMaxK>{});
        });
  }
  auto dw = dw32.to(w.scalar_type());
  at::Tensor db;
  if (has_bias) db = db32.to(w.scalar_type());
  return {dx, dw, db};
}
