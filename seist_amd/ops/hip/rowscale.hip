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

// ---------------------------------------------------------------------------
// Fused residual + DropPath + elementwise Dropout:
//   z = x + pathkeep(n)/pk * dropkeep(i)/dk * y
// One pass replaces bernoulli -> div -> mul (dropout) + the residual
// chain. RNG is the splitmix counter hash on a device-resident seed; the
// kernel snapshots the seed it used into a per-call slot so the backward
// regenerates the same masks under hipGraph replay (where the global
// seed advances every replay).
// ---------------------------------------------------------------------------

unsigned long long* attn_seed_state(const at::Tensor& ref);
void bump_attn_seed(const at::Tensor& ref);

namespace {

__device__ __forceinline__ float rs_rng(unsigned long long seed,
                                        unsigned long long idx) {
  unsigned long long zz = seed + idx * 0x9E3779B97F4A7C15ull;
  zz = (zz ^ (zz >> 30)) * 0xBF58476D1CE4E5B9ull;
  zz = (zz ^ (zz >> 27)) * 0x94D049BB133111EBull;
  zz = zz ^ (zz >> 31);
  return (float)(zz >> 40) * (1.0f / 16777216.0f);
}

template <typename scalar_t, bool PATH, bool DROP>
__global__ void dpd_add_kernel(const scalar_t* __restrict__ x,
                               const scalar_t* __restrict__ y,
                               scalar_t* __restrict__ z,
                               const unsigned long long* __restrict__ gseed,
                               unsigned long long* __restrict__ slot,
                               float path_keep, float inv_pk,
                               float drop_keep, float inv_dk,
                               unsigned long long base, long row_elems,
                               long total) {
  const unsigned long long seed = *gseed;
  if (blockIdx.x == 0 && threadIdx.x == 0) *slot = seed;
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long i = ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
    if (i >= total) return;
    float m = 1.0f;
    if (PATH) {
      const long n = i / row_elems;
      m = rs_rng(seed, base + (unsigned long long)total + n) < path_keep
              ? inv_pk : 0.0f;
    }
    if (DROP && m != 0.0f) {
      m *= rs_rng(seed, base + (unsigned long long)i) < drop_keep
               ? inv_dk : 0.0f;
    }
    z[i] = (scalar_t)((float)x[i] + m * (float)y[i]);
  }
}

template <typename scalar_t, bool PATH, bool DROP>
__global__ void dpd_scale_kernel(const scalar_t* __restrict__ dz,
                                 scalar_t* __restrict__ dy,
                                 const unsigned long long* __restrict__ slot,
                                 float path_keep, float inv_pk,
                                 float drop_keep, float inv_dk,
                                 unsigned long long base, long row_elems,
                                 long total) {
  const unsigned long long seed = *slot;
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long i = ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
    if (i >= total) return;
    float m = 1.0f;
    if (PATH) {
      const long n = i / row_elems;
      m = rs_rng(seed, base + (unsigned long long)total + n) < path_keep
              ? inv_pk : 0.0f;
    }
    if (DROP && m != 0.0f) {
      m *= rs_rng(seed, base + (unsigned long long)i) < drop_keep
               ? inv_dk : 0.0f;
    }
    dy[i] = (scalar_t)(m * (float)dz[i]);
  }
}

}  // namespace

std::vector<at::Tensor> droppath_dropout_add(const at::Tensor& x,
                                             const at::Tensor& y,
                                             double path_p, double drop_p,
                                             long base) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && y.is_contiguous());
  const long total = x.numel();
  const long row_elems = total / x.size(0);
  auto z = at::empty_like(x);
  auto slot = at::empty({1}, x.options().dtype(at::kLong));
  const bool path = path_p > 0.0;
  const bool drop = drop_p > 0.0;
  const float pk = 1.0f - (float)path_p;
  const float dk = 1.0f - (float)drop_p;
  auto stream = at::hip::getCurrentHIPStream();
  unsigned long long* gseed = attn_seed_state(x);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "dpd_add", [&] {
        auto launch = [&](auto p_, auto d_) {
          hipLaunchKernelGGL(
              (dpd_add_kernel<scalar_t, decltype(p_)::value,
                              decltype(d_)::value>),
              dim3(sa::ceil_div(total, (long)kBlock * kEwTile)),
              dim3(kBlock), 0, stream.stream(), x.data_ptr<scalar_t>(),
              y.data_ptr<scalar_t>(), z.data_ptr<scalar_t>(), gseed,
              (unsigned long long*)slot.data_ptr<long>(), pk, 1.0f / pk,
              dk, 1.0f / dk, (unsigned long long)base, row_elems, total);
        };
        if (path) {
          if (drop) launch(std::true_type{}, std::true_type{});
          else launch(std::true_type{}, std::false_type{});
        } else {
          if (drop) launch(std::false_type{}, std::true_type{});
          else launch(std::false_type{}, std::false_type{});
        }
      });
  bump_attn_seed(x);
  return {z, slot};
}

at::Tensor droppath_dropout_scale(const at::Tensor& dz,
                                  const at::Tensor& slot, double path_p,
                                  double drop_p, long base) {
  const long total = dz.numel();
  const long row_elems = total / dz.size(0);
  auto dy = at::empty_like(dz);
  const bool path = path_p > 0.0;
  const bool drop = drop_p > 0.0;
  const float pk = 1.0f - (float)path_p;
  const float dk = 1.0f - (float)drop_p;
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, dz.scalar_type(),
      "dpd_scale", [&] {
        auto launch = [&](auto p_, auto d_) {
          hipLaunchKernelGGL(
              (dpd_scale_kernel<scalar_t, decltype(p_)::value,
                                decltype(d_)::value>),
              dim3(sa::ceil_div(total, (long)kBlock * kEwTile)),
              dim3(kBlock), 0, stream.stream(), dz.data_ptr<scalar_t>(),
              dy.data_ptr<scalar_t>(),
              (const unsigned long long*)slot.data_ptr<long>(), pk,
              1.0f / pk, dk, 1.0f / dk, (unsigned long long)base,
              row_elems, total);
        };
        if (path) {
          if (drop) launch(std::true_type{}, std::true_type{});
          else launch(std::true_type{}, std::false_type{});
        } else {
          if (drop) launch(std::false_type{}, std::true_type{});
          else launch(std::false_type{}, std::false_type{});
        }
      });
  return dy;
}
