// Fused residual + DropPath (stochastic depth): z = x + mask[n] * scale * y
// where mask is a per-sample 0/1 row mask. Replaces the reference-style
// eager chain bernoulli -> div -> mul -> add (models/seist.py DropPath
// sites) with one elementwise pass; backward for y is one row-scale pass
// and dx is the incoming gradient unchanged (no kernel at all).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kEwTile = 4;  // elements per thread (dispatch-rate relief)

template <typename scalar_t, bool HAS_MASK>
__global__ void row_scale_add_kernel(const scalar_t* __restrict__ x,
                                     const scalar_t* __restrict__ y,
                                     const float* __restrict__ mask,
                                     scalar_t* __restrict__ z,
                                     float scale, long row_elems,
                                     long total) {
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long i = ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
    if (i >= total) return;
    float m = scale;
    if (HAS_MASK) m *= mask[i / row_elems];
    z[i] = (scalar_t)((float)x[i] + m * (float)y[i]);
  }
}

template <typename scalar_t, bool HAS_MASK>
__global__ void row_scale_kernel(const scalar_t* __restrict__ y,
                                 const float* __restrict__ mask,
                                 scalar_t* __restrict__ z,
                                 float scale, long row_elems, long total) {
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long i = ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
    if (i >= total) return;
    float m = scale;
    if (HAS_MASK) m *= mask[i / row_elems];
    z[i] = (scalar_t)(m * (float)y[i]);
  }
}

}  // namespace

at::Tensor row_scale_add(const at::Tensor& x, const at::Tensor& y,
                         const c10::optional<at::Tensor>& mask,
                         double scale) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(x.sizes() == y.sizes());
  const long total = x.numel();
  const long row_elems = total / x.size(0);
  auto z = at::empty_like(x);
  const bool has_mask = mask.has_value() && mask->defined();
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "row_scale_add", [&] {
        if (has_mask) {
          hipLaunchKernelGGL((row_scale_add_kernel<scalar_t, true>),
                             dim3(sa::ceil_div(total, (long)kBlock * kEwTile)),
                             dim3(kBlock),
                             0, stream.stream(), x.data_ptr<scalar_t>(),
                             y.data_ptr<scalar_t>(), mask->data_ptr<float>(),
                             z.data_ptr<scalar_t>(), (float)scale, row_elems,
                             total);
        } else {
          hipLaunchKernelGGL((row_scale_add_kernel<scalar_t, false>),
                             dim3(sa::ceil_div(total, (long)kBlock * kEwTile)),
                             dim3(kBlock),
                             0, stream.stream(), x.data_ptr<scalar_t>(),
                             y.data_ptr<scalar_t>(), nullptr,
                             z.data_ptr<scalar_t>(), (float)scale, row_elems,
                             total);
        }
      });
  return z;
}

at::Tensor row_scale(const at::Tensor& y,
                     const c10::optional<at::Tensor>& mask, double scale) {
  TORCH_CHECK(y.is_cuda() && y.is_contiguous());
  const long total = y.numel();
  const long row_elems = total / y.size(0);
  auto z = at::empty_like(y);
  const bool has_mask = mask.has_value() && mask->defined();
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, y.scalar_type(),
      "row_scale", [&] {
        if (has_mask) {
          hipLaunchKernelGGL((row_scale_kernel<scalar_t, true>),
                             dim3(sa::ceil_div(total, (long)kBlock * kEwTile)),
                             dim3(kBlock),
                             0, stream.stream(), y.data_ptr<scalar_t>(),
                             mask->data_ptr<float>(), z.data_ptr<scalar_t>(),
                             (float)scale, row_elems, total);
        } else {
          hipLaunchKernelGGL((row_scale_kernel<scalar_t, false>),
                             dim3(sa::ceil_div(total, (long)kBlock * kEwTile)),
                             dim3(kBlock),
                             0, stream.stream(), y.data_ptr<scalar_t>(),
                             nullptr, z.data_ptr<scalar_t>(),
                             (float)scale, row_elems, total);
        }
      });
  return z;
}
