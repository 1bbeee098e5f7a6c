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
// one block per (co, cig*K), grid-stride over (n, lo); fp32 atomics with
// split over blockIdx.z.
template <typename scalar_t, bool HAS_BIAS>
__global__ void conv1d_dw_kernel(const scalar_t* __restrict__ dy,
                                 const scalar_t* __restrict__ x,
                                 float* __restrict__ dw,
                                 float* __restrict__ db,
                                 int N, int Ci, int Co, long L, long Lo,
                                 int K, int stride, int padl, int dil,
                                 int G, int nsplit) {
  __shared__ float red[kBlock / sa::kWave];

  const int co = blockIdx.x;
  const int wi = blockIdx.y;  // cig * K + k
  const int cig = wi / K;
  const int k = wi - cig * K;
  const int Cig = Ci / G;
  const int g = co / (Co / G);
  const int ci = g * Cig + cig;

  const long total = (long)N * Lo;
  const long chunk = (total + nsplit - 1) / nsplit;
  const long k0 = (long)blockIdx.z * chunk;
  const long k1 = min(total, k0 + chunk);

  float acc = 0.0f;
  float bacc = 0.0f;
  for (long t = k0 + threadIdx.x; t < k1; t += kBlock) {
    const long n = t / Lo;
    const long lo = t - n * Lo;
    const float dyv = (float)dy[((long)n * Co + co) * Lo + lo];
    const long li = lo * stride - padl + (long)k * dil;
    if (li >= 0 && li < L) {
      acc += dyv * (float)x[((long)n * Ci + ci) * L + li];
    }
    if (HAS_BIAS && wi == 0) bacc += dyv;
  }
  acc = sa::block_reduce_sum(acc, red);
  if (threadIdx.x == 0) atomicAdd(&dw[((long)co * Cig + cig) * K + k], acc);
  if (HAS_BIAS && wi == 0) {
    __syncthreads();
    bacc = sa::block_reduce_sum(bacc, red);
    if (threadIdx.x == 0) atomicAdd(&db[co], bacc);
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
    const int nsplit = std::max(
        1, std::min<int>(32, (int)(((long)N * Lo) / 65536) + 1));
    dim3 grid(Co, Cig * K, nsplit);
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
        "conv1d_dw", [&] {
          if (has_bias) {
            hipLaunchKernelGGL((conv1d_dw_kernel<scalar_t, true>), grid,
                               dim3(kBlock), 0, stream.stream(),
                               dy.data_ptr<scalar_t>(), x.data_ptr<scalar_t>(),
                               dw32.data_ptr<float>(), db32.data_ptr<float>(),
                               N, Ci, Co, L, Lo, K, (int)stride, (int)padl,
                               (int)dilation, (int)groups, nsplit);
          } else {
            hipLaunchKernelGGL((conv1d_dw_kernel<scalar_t, false>), grid,
                               dim3(kBlock), 0, stream.stream(),
                               dy.data_ptr<scalar_t>(), x.data_ptr<scalar_t>(),
                               dw32.data_ptr<float>(), nullptr,
                               N, Ci, Co, L, Lo, K, (int)stride, (int)padl,
                               (int)dilation, (int)groups, nsplit);
          }
        });
  }
  auto dw = dw32.to(w.scalar_type());
  at::Tensor db;
  if (has_bias) db = db32.to(w.scalar_type());
  return {dx, dw, db};
}
