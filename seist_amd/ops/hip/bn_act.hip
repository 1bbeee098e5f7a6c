// Fused BatchNorm1d (+GELU/ReLU) over (N, C, L) — K7 + K14 of
// SURVEY.md §2.4. Replaces the reference's norm_layer+act pairs
// (models/seist.py:145-154, 223-224, phasenet/eqtransformer BNs).
//
// Stats, parameters and running buffers are fp32 regardless of activation
// dtype. Training forward = one split-reduction kernel (atomics into a
// per-channel fp32 scratch) + one finalize kernel + one fused
// normalize+activation pass; eval forward is a single pass. Backward
// mirrors that (reduce dgamma/dbeta, then one elementwise dx pass).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

namespace {

constexpr int kBlock = 256;

// Channel-routed row lookup for the virtual channel-concat forms: the BN
// runs over cat(x0, x1[, x2]) without the cat existing (MSMC/MPT sites,
// reference models/seist.py:308-318, 486-504). cb1 == C degenerates to
// the single-tensor form.
template <typename T>
__device__ __forceinline__ const T* bn_row(const T* x0, const T* x1,
                                           const T* x2, int cb1, int cb2,
                                           int C, long n, int c, long L) {
  if (c < cb1) return x0 + (n * cb1 + c) * L;
  if (c < cb2) return x1 + (n * (long)(cb2 - cb1) + (c - cb1)) * L;
  return x2 + (n * (long)(C - cb2) + (c - cb2)) * L;
}

template <typename T>
__device__ __forceinline__ T* bn_row_mut(T* x0, T* x1, T* x2, int cb1,
                                         int cb2, int C, long n, int c,
                                         long L) {
  if (c < cb1) return x0 + (n * cb1 + c) * L;
  if (c < cb2) return x1 + (n * (long)(cb2 - cb1) + (c - cb1)) * L;
  return x2 + (n * (long)(C - cb2) + (c - cb2)) * L;
}


template <typename scalar_t, bool MULTI = false>
__global__ void bn_sums_kernel(const scalar_t* __restrict__ x,
                               float* __restrict__ part,  // (C, nsplit, 2)
                               int C, long NL, long L,
                               const scalar_t* x1 = nullptr,
                               const scalar_t* x2 = nullptr,
                               int cb1 = 0, int cb2 = 0) {
  __shared__ float red[kBlock / sa::kWave];
  const int c = blockIdx.x;
  const int split = blockIdx.y;
  const int nsplit = gridDim.y;
  const long N = NL / L;
  const long nchunk = (N + nsplit - 1) / nsplit;
  const long n0 = (long)split * nchunk;
  const long n1 = min(N, n0 + nchunk);

  float s = 0.0f, s2 = 0.0f;
  for (long n = n0; n < n1; ++n) {
    const scalar_t* xr = MULTI ? bn_row(x, x1, x2, cb1, cb2, C, n, c, L)
                               : x + (n * C + c) * L;
    for (long l = threadIdx.x; l < L; l += kBlock) {
      const float v = (float)xr[l];
      s += v;
      s2 += v * v;
    }
  }
  s = sa::block_reduce_sum(s, red);
  __syncthreads();
  s2 = sa::block_reduce_sum(s2, red);
  if (threadIdx.x == 0) {
    part[((long)c * nsplit + split) * 2 + 0] = s;
    part[((long)c * nsplit + split) * 2 + 1] = s2;
  }
}

// one wave per channel: the per-channel partial reduction runs across the
// 64 lanes instead of as one serial dependent-load loop (the thread-per-
// channel version was latency-bound at ~12 us for KB-scale reads)
// CMAJOR: part layout (C, nsplit, 2); else split-major (nsplit, C, 2)
// (the conv-epilogue slabs are split-major so each block flushes one
// coalesced run). The two coincide at nsplit == 1.
__global__ void bn_finalize_kernel(const float* __restrict__ part,
                                   int nsplit, int cmajor,
                                   float* __restrict__ mean,
                                   float* __restrict__ invstd,
                                   float* __restrict__ running_mean,
                                   float* __restrict__ running_var,
                                   int C, long NL, float momentum,
                                   float eps) {
  const int lane = threadIdx.x & (sa::kWave - 1);
  const int c = blockIdx.x * (blockDim.x / sa::kWave)
                + threadIdx.x / sa::kWave;
  if (c >= C) return;
  float s = 0.0f, s2 = 0.0f;
  for (int j = lane; j < nsplit; j += sa::kWave) {
    const long o = cmajor ? ((long)c * nsplit + j) : ((long)j * C + c);
    s += part[o * 2 + 0];
    s2 += part[o * 2 + 1];
  }
  s = sa::warp_reduce_sum(s);
  s2 = sa::warp_reduce_sum(s2);
  if (lane != 0) return;
  const float m = s / NL;
  const float var = fmaxf(s2 / NL - m * m, 0.0f);
  mean[c] = m;
  invstd[c] = rsqrtf(var + eps);
  if (running_mean != nullptr) {
    const float unbiased = var * ((float)NL / fmaxf((float)(NL - 1), 1.0f));
    running_mean[c] = (1.0f - momentum) * running_mean[c] + momentum * m;
    running_var[c] = (1.0f - momentum) * running_var[c] + momentum * unbiased;
  }
}

// 4 elements per thread: elementwise grids of 128k one-wave blocks were
// dispatch-rate bound, not bandwidth bound
constexpr int kEwTile = 4;

template <typename scalar_t, bool MULTI = false>
__global__ void bn_apply_kernel(const scalar_t* __restrict__ x,
                                scalar_t* __restrict__ y,
                                const float* __restrict__ mean,
                                const float* __restrict__ invstd,
                                const float* __restrict__ gamma,
                                const float* __restrict__ beta,
                                int C, long L, long total, int act,
                                const scalar_t* x1 = nullptr,
                                const scalar_t* x2 = nullptr,
                                int cb1 = 0, int cb2 = 0) {
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long i =
        ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
    if (i >= total) return;
    const long row = i / L;
    const int c = (int)(row % C);
    float xv;
    if (MULTI) {
      xv = (float)bn_row(x, x1, x2, cb1, cb2, C, row / C, c, L)[i - row * L];
    } else {
      xv = (float)x[i];
    }
    const float xh = (xv - mean[c]) * invstd[c];
    const float pre = xh * gamma[c] + beta[c];
    y[i] = (scalar_t)sa::act_fwd(pre, act);
  }
}

template <typename scalar_t, bool MULTI = false>
__global__ void bn_bwd_sums_kernel(const scalar_t* __restrict__ dy,
                                   const scalar_t* __restrict__ x,
                                   const float* __restrict__ mean,
                                   const float* __restrict__ invstd,
                                   const float* __restrict__ gamma,
                                   const float* __restrict__ beta,
                                   float* __restrict__ part,  // (C,nsplit,2)
                                   int C, long NL, long L, int act,
                                   const scalar_t* x1 = nullptr,
                                   const scalar_t* x2 = nullptr,
                                   int cb1 = 0, int cb2 = 0) {
  __shared__ float red[kBlock / sa::kWave];
  const int c = blockIdx.x;
  const int split = blockIdx.y;
  const int nsplit = gridDim.y;
  const long N = NL / L;
  const long nchunk = (N + nsplit - 1) / nsplit;
  const long n0 = (long)split * nchunk;
  const long n1 = min(N, n0 + nchunk);

  const float m = mean[c], is = invstd[c], g = gamma[c], b = beta[c];
  float s1 = 0.0f, s2 = 0.0f;
  for (long n = n0; n < n1; ++n) {
    const scalar_t* xr = MULTI ? bn_row(x, x1, x2, cb1, cb2, C, n, c, L)
                               : x + (n * C + c) * L;
    const scalar_t* dyr = dy + (n * C + c) * L;
    for (long l = threadIdx.x; l < L; l += kBlock) {
      const float xh = ((float)xr[l] - m) * is;
      float d = (float)dyr[l];
      if (act != sa::ACT_NONE) d *= sa::act_grad(xh * g + b, act);
      s1 += d;
      s2 += d * xh;
    }
  }
  s1 = sa::block_reduce_sum(s1, red);
  __syncthreads();
  s2 = sa::block_reduce_sum(s2, red);
  if (threadIdx.x == 0) {
    part[((long)c * nsplit + split) * 2 + 0] = s1;
    part[((long)c * nsplit + split) * 2 + 1] = s2;
  }
}

// reduce (C, nsplit, 2) partials -> (C, 2); one wave per channel
__global__ void bn_part_reduce_kernel(const float* __restrict__ part,
                                      float* __restrict__ out,
                                      int C, int nsplit, int cmajor) {
  const int lane = threadIdx.x & (sa::kWave - 1);
  const int c = blockIdx.x * (blockDim.x / sa::kWave)
                + threadIdx.x / sa::kWave;
  if (c >= C) return;
  float s1 = 0.0f, s2 = 0.0f;
  for (int j = lane; j < nsplit; j += sa::kWave) {
    const long o = cmajor ? ((long)c * nsplit + j) : ((long)j * C + c);
    s1 += part[o * 2 + 0];
    s2 += part[o * 2 + 1];
  }
  s1 = sa::warp_reduce_sum(s1);
  s2 = sa::warp_reduce_sum(s2);
  if (lane != 0) return;
  out[c * 2 + 0] = s1;
  out[c * 2 + 1] = s2;
}

template <typename scalar_t, bool TRAINING, bool MULTI = false>
__global__ void bn_bwd_dx_kernel(const scalar_t* __restrict__ dy,
                                 const scalar_t* __restrict__ x,
                                 scalar_t* __restrict__ dx,
                                 const float* __restrict__ mean,
                                 const float* __restrict__ invstd,
                                 const float* __restrict__ gamma,
                                 const float* __restrict__ beta,
                                 const float* __restrict__ sums,  // dbeta,dgamma
                                 int C, long L, long total, long NL, int act,
                                 const scalar_t* x1 = nullptr,
                                 const scalar_t* x2 = nullptr,
                                 scalar_t* dx1 = nullptr,
                                 scalar_t* dx2 = nullptr,
                                 int cb1 = 0, int cb2 = 0) {
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long i =
        ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
    if (i >= total) return;
    const long row = i / L;
    const int c = (int)(row % C);
    const long l = i - row * L;
    const float m = mean[c], is = invstd[c], g = gamma[c], b = beta[c];
    const float xv = MULTI
        ? (float)bn_row(x, x1, x2, cb1, cb2, C, row / C, c, L)[l]
        : (float)x[i];
    const float xh = (xv - m) * is;
    float d = (float)dy[i];
    if (act != sa::ACT_NONE) d *= sa::act_grad(xh * g + b, act);
    float v;
    if (TRAINING) {
      const float dbeta = sums[c * 2 + 0];
      const float dgamma = sums[c * 2 + 1];
      v = (g * is / (float)NL) * ((float)NL * d - dbeta - xh * dgamma);
    } else {
      v = d * g * is;
    }
    if (MULTI) {
      bn_row_mut(dx, dx1, dx2, cb1, cb2, C, row / C, c, L)[l] = (scalar_t)v;
    } else {
      dx[i] = (scalar_t)v;
    }
  }
}

int pick_nsplit(long N, int C) {
  // enough blocks to fill 256 CUs several times over, bounded by N
  const int want = std::max(1, 2048 / std::max(C, 1));
  return std::max(1, std::min<int>((int)N, want));
}

}  // namespace

std::vector<at::Tensor> bn_act_fwd(const at::Tensor& x, const at::Tensor& gamma,
                                   const at::Tensor& beta,
                                   const c10::optional<at::Tensor>& running_mean,
                                   const c10::optional<at::Tensor>& running_var,
                                   bool training, double momentum, double eps,
                                   long act) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  const int N = x.size(0), C = x.size(1);
  const long L = x.size(2);
  const long NL = (long)N * L;
  const long total = (long)N * C * L;
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = x.options().dtype(at::kFloat);

  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();

  at::Tensor mean, invstd;
  const bool has_running = running_mean.has_value() && running_mean->defined();
  if (training) {
    const int nsplit = pick_nsplit(N, C);
    auto part = at::empty({C, nsplit, 2}, opts);
    mean = at::empty({C}, opts);
    invstd = at::empty({C}, opts);
    TORCH_CHECK(!has_running || running_mean->scalar_type() == at::kFloat,
                "running stats must be fp32");
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
        "bn_sums", [&] {
          hipLaunchKernelGGL((bn_sums_kernel<scalar_t>),
                             dim3(C, nsplit), dim3(kBlock), 0,
                             stream.stream(), x.data_ptr<scalar_t>(),
                             part.data_ptr<float>(), C, NL, L);
        });
    hipLaunchKernelGGL(bn_finalize_kernel, dim3(sa::ceil_div(C, 4)),
                       dim3(256), 0, stream.stream(),
                       part.data_ptr<float>(), nsplit, 1,
                       mean.data_ptr<float>(),
                       invstd.data_ptr<float>(),
                       has_running ? running_mean->data_ptr<float>() : nullptr,
                       has_running ? running_var->data_ptr<float>() : nullptr,
                       C, NL, (float)momentum, (float)eps);
  } else {
    TORCH_CHECK(has_running, "eval mode requires running stats");
    mean = running_mean->to(at::kFloat).contiguous();
    invstd = at::rsqrt(running_var->to(at::kFloat) + eps).contiguous();
  }

  auto y = at::empty_like(x);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "bn_apply", [&] {
        hipLaunchKernelGGL((bn_apply_kernel<scalar_t>),
                           dim3(sa::ceil_div(total, (long)kBlock * kEwTile)),
                           dim3(kBlock), 0,
                           stream.stream(), x.data_ptr<scalar_t>(),
                           y.data_ptr<scalar_t>(), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), g32.data_ptr<float>(),
                           b32.data_ptr<float>(), C, L, total, (int)act);
      });
  return {y, mean, invstd};
}

// --- SyncBN split form -----------------------------------
// The cross-rank form runs the same kernels with an RCCL all-reduce of the
// tiny (C, 2) sums tensor between the local reduction and the finalize —
// the native BN kernel stays in use under distributed training (the
// reference converts to torch SyncBatchNorm and loses its fused path,
// train.py:374).

at::Tensor bn_sums_only(const at::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  const int N = x.size(0), C = x.size(1);
  const long L = x.size(2);
  const long NL = (long)N * L;
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = x.options().dtype(at::kFloat);
  const int nsplit = pick_nsplit(N, C);
  auto part = at::empty({C, nsplit, 2}, opts);
  auto sums = at::empty({C, 2}, opts);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "bn_sums", [&] {
        hipLaunchKernelGGL((bn_sums_kernel<scalar_t>), dim3(C, nsplit),
                           dim3(kBlock), 0, stream.stream(),
                           x.data_ptr<scalar_t>(), part.data_ptr<float>(),
                           C, NL, L);
      });
  hipLaunchKernelGGL(bn_part_reduce_kernel, dim3(sa::ceil_div(C, 4)),
                     dim3(256), 0, stream.stream(), part.data_ptr<float>(),
                     sums.data_ptr<float>(), C, nsplit, 1);
  return sums;
}

std::vector<at::Tensor> bn_act_fwd_from_sums(
    const at::Tensor& x, const at::Tensor& sums, double count,
    const at::Tensor& gamma, const at::Tensor& beta,
    const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, double momentum, double eps,
    long act) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  TORCH_CHECK(sums.is_cuda() && sums.is_contiguous());
  const int N = x.size(0), C = x.size(1);
  const long L = x.size(2);
  const long total = (long)N * C * L;
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = x.options().dtype(at::kFloat);
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();
  auto mean = at::empty({C}, opts);
  auto invstd = at::empty({C}, opts);
  const bool has_running = running_mean.has_value() && running_mean->defined();
  // sums is (C, 2) == part layout with nsplit == 1
  hipLaunchKernelGGL(bn_finalize_kernel, dim3(sa::ceil_div(C, 4)), dim3(256),
                     0, stream.stream(), sums.data_ptr<float>(), 1, 1,
                     mean.data_ptr<float>(), invstd.data_ptr<float>(),
                     has_running ? running_mean->data_ptr<float>() : nullptr,
                     has_running ? running_var->data_ptr<float>() : nullptr,
                     C, (long)count, (float)momentum, (float)eps);
  auto y = at::empty_like(x);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "bn_apply", [&] {
        hipLaunchKernelGGL((bn_apply_kernel<scalar_t>),
                           dim3(sa::ceil_div(total, (long)kBlock * kEwTile)),
                           dim3(kBlock), 0, stream.stream(),
                           x.data_ptr<scalar_t>(), y.data_ptr<scalar_t>(),
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           g32.data_ptr<float>(), b32.data_ptr<float>(),
                           C, L, total, (int)act);
      });
  return {y, mean, invstd};
}

// fwd consuming a producer-collected (C, nsplit, 2) partial slab
// (fusion step 1): no bn_sums pass over x.
std::vector<at::Tensor> bn_act_fwd_with_part(
    const at::Tensor& x, const at::Tensor& part, const at::Tensor& gamma,
    const at::Tensor& beta, const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, double momentum, double eps,
    long act) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  TORCH_CHECK(part.is_contiguous() && part.dim() == 3
              && part.size(1) == x.size(1) && part.size(2) == 2);
  const int N = x.size(0), C = x.size(1);
  const long L = x.size(2);
  const long NL = (long)N * L;
  const long total = (long)N * C * L;
  const int nsplit = part.size(0);
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = x.options().dtype(at::kFloat);
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();
  auto mean = at::empty({C}, opts);
  auto invstd = at::empty({C}, opts);
  const bool has_running = running_mean.has_value() && running_mean->defined();
  hipLaunchKernelGGL(bn_finalize_kernel, dim3(sa::ceil_div(C, 4)), dim3(256),
                     0, stream.stream(), part.data_ptr<float>(), nsplit, 0,
                     mean.data_ptr<float>(), invstd.data_ptr<float>(),
                     has_running ? running_mean->data_ptr<float>() : nullptr,
                     has_running ? running_var->data_ptr<float>() : nullptr,
                     C, NL, (float)momentum, (float)eps);
  auto y = at::empty_like(x);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "bn_apply", [&] {
        hipLaunchKernelGGL((bn_apply_kernel<scalar_t>),
                           dim3(sa::ceil_div(total, (long)kBlock * kEwTile)),
                           dim3(kBlock), 0, stream.stream(),
                           x.data_ptr<scalar_t>(), y.data_ptr<scalar_t>(),
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           g32.data_ptr<float>(), b32.data_ptr<float>(),
                           C, L, total, (int)act);
      });
  return {y, mean, invstd};
}

// reduce a (C, nsplit, 2) slab to (C, 2) — used by the SyncBN path to
// all-reduce producer-collected partials
at::Tensor bn_part_to_sums(const at::Tensor& part) {
  // split-major producer slab (nsplit, C, 2)
  const int C = part.size(1);
  const int nsplit = part.size(0);
  auto sums = at::empty({C, 2}, part.options());
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(bn_part_reduce_kernel, dim3(sa::ceil_div(C, 4)),
                     dim3(256), 0, stream.stream(), part.data_ptr<float>(),
                     sums.data_ptr<float>(), C, nsplit, 0);
  return sums;
}

// finalize-only: mean/invstd (+ running update) from (C, 2) sums —
// used by the BN+act+conv fused op, which never materializes the apply
std::vector<at::Tensor> bn_finalize_only(
    const at::Tensor& sums, double count,
    const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, double momentum,
    double eps) {
  const int C = sums.size(0);
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = sums.options();
  auto mean = at::empty({C}, opts);
  auto invstd = at::empty({C}, opts);
  const bool has_running = running_mean.has_value() && running_mean->defined();
  hipLaunchKernelGGL(bn_finalize_kernel, dim3(sa::ceil_div(C, 4)), dim3(256),
                     0, stream.stream(), sums.data_ptr<float>(), 1, 1,
                     mean.data_ptr<float>(), invstd.data_ptr<float>(),
                     has_running ? running_mean->data_ptr<float>() : nullptr,
                     has_running ? running_var->data_ptr<float>() : nullptr,
                     C, (long)count, (float)momentum, (float)eps);
  return {mean, invstd};
}

// eval-mode dx only (dx = act_grad * gamma * invstd * dy): the act-only
// fused conv's backward — no dgamma/dbeta reductions
at::Tensor bn_bwd_dx_eval(const at::Tensor& dy, const at::Tensor& x,
                          const at::Tensor& mean, const at::Tensor& invstd,
                          const at::Tensor& gamma, const at::Tensor& beta,
                          long act) {
  const int N = x.size(0), C = x.size(1);
  const long L = x.size(2);
  const long total = (long)N * C * L;
  auto stream = at::hip::getCurrentHIPStream();
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();
  auto dx = at::empty_like(x);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "bn_bwd_dx_eval", [&] {
        hipLaunchKernelGGL((bn_bwd_dx_kernel<scalar_t, false>),
                           dim3(sa::ceil_div(total, (long)kBlock * kEwTile)),
                           dim3(kBlock), 0, stream.stream(),
                           dy.data_ptr<scalar_t>(), x.data_ptr<scalar_t>(),
                           dx.data_ptr<scalar_t>(), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), g32.data_ptr<float>(),
                           b32.data_ptr<float>(), nullptr, C, L, total,
                           (long)N * L, (int)act);
      });
  return dx;
}

at::Tensor bn_bwd_sums_only(const at::Tensor& dy, const at::Tensor& x,
                            const at::Tensor& mean, const at::Tensor& invstd,
                            const at::Tensor& gamma, const at::Tensor& beta,
                            long act) {
  const int N = x.size(0), C = x.size(1);
  const long L = x.size(2);
  const long NL = (long)N * L;
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = x.options().dtype(at::kFloat);
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();
  const int nsplit = pick_nsplit(N, C);
  auto part = at::empty({C, nsplit, 2}, opts);
  auto sums = at::empty({C, 2}, opts);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "bn_bwd_sums", [&] {
        hipLaunchKernelGGL((bn_bwd_sums_kernel<scalar_t>), dim3(C, nsplit),
                           dim3(kBlock), 0, stream.stream(),
                           dy.data_ptr<scalar_t>(), x.data_ptr<scalar_t>(),
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           g32.data_ptr<float>(), b32.data_ptr<float>(),
                           part.data_ptr<float>(), C, NL, L, (int)act);
      });
  hipLaunchKernelGGL(bn_part_reduce_kernel, dim3(sa::ceil_div(C, 4)),
                     dim3(256), 0, stream.stream(), part.data_ptr<float>(),
                     sums.data_ptr<float>(), C, nsplit, 1);
  return sums;
}

at::Tensor bn_bwd_dx_from_sums(const at::Tensor& dy, const at::Tensor& x,
                               const at::Tensor& mean,
                               const at::Tensor& invstd,
                               const at::Tensor& gamma, const at::Tensor& beta,
                               const at::Tensor& sums, double count,
                               long act) {
  const int N = x.size(0), C = x.size(1);
  const long L = x.size(2);
  const long total = (long)N * C * L;
  auto stream = at::hip::getCurrentHIPStream();
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();
  auto dx = at::empty_like(x);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "bn_bwd_dx", [&] {
        hipLaunchKernelGGL((bn_bwd_dx_kernel<scalar_t, true>),
                           dim3(sa::ceil_div(total, (long)kBlock * kEwTile)),
                           dim3(kBlock), 0, stream.stream(),
                           dy.data_ptr<scalar_t>(), x.data_ptr<scalar_t>(),
                           dx.data_ptr<scalar_t>(), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), g32.data_ptr<float>(),
                           b32.data_ptr<float>(), sums.data_ptr<float>(),
                           C, L, total, (long)count, (int)act);
      });
  return dx;
}

std::vector<at::Tensor> bn_act_bwd(const at::Tensor& dy, const at::Tensor& x,
                                   const at::Tensor& gamma,
                                   const at::Tensor& beta,
                                   const at::Tensor& mean,
                                   const at::Tensor& invstd, bool training,
                                   long act) {
  const int N = x.size(0), C = x.size(1);
  const long L = x.size(2);
  const long NL = (long)N * L;
  const long total = (long)N * C * L;
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = x.options().dtype(at::kFloat);

  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();

  const int nsplit = pick_nsplit(N, C);
  auto part = at::empty({C, nsplit, 2}, opts);
  auto sums = at::empty({C, 2}, opts);  // dbeta, dgamma
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "bn_bwd_sums", [&] {
        hipLaunchKernelGGL((bn_bwd_sums_kernel<scalar_t>),
                           dim3(C, nsplit), dim3(kBlock), 0,
                           stream.stream(), dy.data_ptr<scalar_t>(),
                           x.data_ptr<scalar_t>(), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), g32.data_ptr<float>(),
                           b32.data_ptr<float>(), part.data_ptr<float>(),
                           C, NL, L, (int)act);
      });
  hipLaunchKernelGGL(bn_part_reduce_kernel, dim3(sa::ceil_div(C, 4)),
                     dim3(256), 0, stream.stream(), part.data_ptr<float>(),
                     sums.data_ptr<float>(), C, nsplit, 1);

  auto dx = at::empty_like(x);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "bn_bwd_dx", [&] {
        if (training) {
          hipLaunchKernelGGL((bn_bwd_dx_kernel<scalar_t, true>),
                             dim3(sa::ceil_div(total,
                                               (long)kBlock * kEwTile)),
                             dim3(kBlock), 0, stream.stream(), dy.data_ptr<scalar_t>(),
                             x.data_ptr<scalar_t>(), dx.data_ptr<scalar_t>(),
                             mean.data_ptr<float>(), invstd.data_ptr<float>(),
                             g32.data_ptr<float>(), b32.data_ptr<float>(),
                             sums.data_ptr<float>(), C, L, total, NL,
                             (int)act);
        } else {
          hipLaunchKernelGGL((bn_bwd_dx_kernel<scalar_t, false>),
                             dim3(sa::ceil_div(total,
                                               (long)kBlock * kEwTile)),
                             dim3(kBlock), 0, stream.stream(), dy.data_ptr<scalar_t>(),
                             x.data_ptr<scalar_t>(), dx.data_ptr<scalar_t>(),
                             mean.data_ptr<float>(), invstd.data_ptr<float>(),
                             g32.data_ptr<float>(), b32.data_ptr<float>(),
                             sums.data_ptr<float>(), C, L, total, NL,
                             (int)act);
        }
      });

  auto split = sums.unbind(1);
  auto dbeta = split[0].to(gamma.scalar_type());
  auto dgamma = split[1].to(gamma.scalar_type());
  return {dx, dgamma, dbeta};
}

// ---------------------------------------------------------------------------
// Virtual channel-concat forms: BN(+act) over cat(xs) without the cat
// (MSMC/MPT concat->norm sites). Backward writes per-input contiguous
// gradients, so the cat's backward narrow+copy pass disappears too.
// ---------------------------------------------------------------------------

std::vector<at::Tensor> bn_act_cat_fwd(
    std::vector<at::Tensor> xs, const at::Tensor& gamma,
    const at::Tensor& beta, const c10::optional<at::Tensor>& running_mean,
    const c10::optional<at::Tensor>& running_var, bool training,
    double momentum, double eps, long act) {
  TORCH_CHECK(xs.size() >= 2 && xs.size() <= 3);
  const int N = xs[0].size(0);
  const long L = xs[0].size(2);
  int C = 0;
  for (auto& x : xs) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.size(0) == N
                && x.size(2) == L);
    C += x.size(1);
  }
  const int cb1 = xs[0].size(1);
  const int cb2 = cb1 + xs[1].size(1);
  const long NL = (long)N * L;
  const long total = (long)N * C * L;
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = xs[0].options().dtype(at::kFloat);
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();

  at::Tensor mean, invstd;
  const bool has_running = running_mean.has_value() && running_mean->defined();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, xs[0].scalar_type(),
      "bn_cat", [&] {
        const scalar_t* x0 = xs[0].data_ptr<scalar_t>();
        const scalar_t* x1 = xs[1].data_ptr<scalar_t>();
        const scalar_t* x2 = xs.size() > 2 ? xs[2].data_ptr<scalar_t>() : x1;
        if (training) {
          const int nsplit = pick_nsplit(N, C);
          auto part = at::empty({C, nsplit, 2}, opts);
          mean = at::empty({C}, opts);
          invstd = at::empty({C}, opts);
          hipLaunchKernelGGL((bn_sums_kernel<scalar_t, true>),
                             dim3(C, nsplit), dim3(kBlock), 0,
                             stream.stream(), x0, part.data_ptr<float>(),
                             C, NL, L, x1, x2, cb1,
                             xs.size() > 2 ? cb2 : C);
          hipLaunchKernelGGL(bn_finalize_kernel, dim3(sa::ceil_div(C, 4)),
                             dim3(256), 0, stream.stream(),
                             part.data_ptr<float>(), nsplit, 1,
                             mean.data_ptr<float>(), invstd.data_ptr<float>(),
                             has_running ? running_mean->data_ptr<float>()
                                         : nullptr,
                             has_running ? running_var->data_ptr<float>()
                                         : nullptr,
                             C, NL, (float)momentum, (float)eps);
        } else {
          TORCH_CHECK(has_running, "eval mode requires running stats");
          mean = running_mean->to(at::kFloat).contiguous();
          invstd = at::rsqrt(running_var->to(at::kFloat) + eps).contiguous();
        }
        auto y = at::empty({N, C, L}, xs[0].options());
        hipLaunchKernelGGL((bn_apply_kernel<scalar_t, true>),
                           dim3(sa::ceil_div(total, (long)kBlock * kEwTile)),
                           dim3(kBlock), 0, stream.stream(), x0,
                           y.data_ptr<scalar_t>(), mean.data_ptr<float>(),
                           invstd.data_ptr<float>(), g32.data_ptr<float>(),
                           b32.data_ptr<float>(), C, L, total, (int)act,
                           x1, x2, cb1, xs.size() > 2 ? cb2 : C);
        mean = mean;  // keep in scope
        invstd = invstd;
        xs[0] = y;  // reuse slot 0 to smuggle y out of the dispatch lambda
      });
  return {xs[0], mean, invstd};
}

std::vector<at::Tensor> bn_act_cat_bwd(
    const at::Tensor& dy, std::vector<at::Tensor> xs,
    const at::Tensor& gamma, const at::Tensor& beta, const at::Tensor& mean,
    const at::Tensor& invstd, bool training, long act) {
  const int N = xs[0].size(0);
  const long L = xs[0].size(2);
  int C = 0;
  for (auto& x : xs) C += x.size(1);
  const int cb1 = xs[0].size(1);
  const int cb2 = cb1 + xs[1].size(1);
  const long NL = (long)N * L;
  const long total = (long)N * C * L;
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = xs[0].options().dtype(at::kFloat);
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();

  const int nsplit = pick_nsplit(N, C);
  auto part = at::empty({C, nsplit, 2}, opts);
  auto sums = at::empty({C, 2}, opts);
  std::vector<at::Tensor> dxs;
  for (auto& x : xs) dxs.push_back(at::empty_like(x));
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, xs[0].scalar_type(),
      "bn_cat_bwd", [&] {
        const scalar_t* x0 = xs[0].data_ptr<scalar_t>();
        const scalar_t* x1 = xs[1].data_ptr<scalar_t>();
        const scalar_t* x2 = xs.size() > 2 ? xs[2].data_ptr<scalar_t>() : x1;
        scalar_t* d0 = dxs[0].data_ptr<scalar_t>();
        scalar_t* d1 = dxs[1].data_ptr<scalar_t>();
        scalar_t* d2 = xs.size() > 2 ? dxs[2].data_ptr<scalar_t>() : d1;
        const int b2 = xs.size() > 2 ? cb2 : C;
        hipLaunchKernelGGL((bn_bwd_sums_kernel<scalar_t, true>),
                           dim3(C, nsplit), dim3(kBlock), 0, stream.stream(),
                           dy.data_ptr<scalar_t>(), x0,
                           mean.data_ptr<float>(), invstd.data_ptr<float>(),
                           g32.data_ptr<float>(), b32.data_ptr<float>(),
                           part.data_ptr<float>(), C, NL, L, (int)act,
                           x1, x2, cb1, b2);
        hipLaunchKernelGGL(bn_part_reduce_kernel, dim3(sa::ceil_div(C, 4)),
                           dim3(256), 0, stream.stream(),
                           part.data_ptr<float>(), sums.data_ptr<float>(),
                           C, nsplit, 1);
        if (training) {
          hipLaunchKernelGGL((bn_bwd_dx_kernel<scalar_t, true, true>),
                             dim3(sa::ceil_div(total,
                                               (long)kBlock * kEwTile)),
                             dim3(kBlock), 0, stream.stream(),
                             dy.data_ptr<scalar_t>(), x0, d0,
                             mean.data_ptr<float>(), invstd.data_ptr<float>(),
                             g32.data_ptr<float>(), b32.data_ptr<float>(),
                             sums.data_ptr<float>(), C, L, total, NL,
                             (int)act, x1, x2, d1, d2, cb1, b2);
        } else {
          hipLaunchKernelGGL((bn_bwd_dx_kernel<scalar_t, false, true>),
                             dim3(sa::ceil_div(total,
                                               (long)kBlock * kEwTile)),
                             dim3(kBlock), 0, stream.stream(),
                             dy.data_ptr<scalar_t>(), x0, d0,
                             mean.data_ptr<float>(), invstd.data_ptr<float>(),
                             g32.data_ptr<float>(), b32.data_ptr<float>(),
                             sums.data_ptr<float>(), C, L, total, NL,
                             (int)act, x1, x2, d1, d2, cb1, b2);
        }
      });
  auto split = sums.unbind(1);
  auto dbeta = split[0].to(gamma.scalar_type());
  auto dgamma = split[1].to(gamma.scalar_type());
  std::vector<at::Tensor> outs = dxs;
  outs.push_back(dgamma);
  outs.push_back(dbeta);
  return outs;
}
