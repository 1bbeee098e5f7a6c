// Fused avg+max pooling (ceil mode) and 1D linear interpolation —
// K12/K13 of SURVEY.md §2.4. avg+max is the LocalAwareAggregation
// primitive (reference models/seist.py:80-93: AvgPool1d + MaxPool1d
// summed — two kernels + an add in eager; one pass here). Linear
// interpolation matches F.interpolate(mode='linear', align_corners=False)
// (reference models/seist.py:563-566 upsampling head).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kEwTile = 4;  // elements per thread (dispatch-rate relief)

template <typename scalar_t>
__global__ void avgmax_fwd_kernel(const scalar_t* __restrict__ x,
                                  scalar_t* __restrict__ y,
                                  int* __restrict__ argmax,
                                  long L, long Lo, int k, long rows) {
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
  const long i = ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
  if (i >= rows * Lo) return;
  const long row = i / Lo;
  const long lo = i - row * Lo;
  const long lo0 = lo * k;
  const long lo1 = min(lo0 + (long)k, L);
  const scalar_t* xr = x + row * L;
  float s = 0.0f;
  float mx = -INFINITY;
  int mi = (int)lo0;
  for (long l = lo0; l < lo1; ++l) {
    const float v = (float)xr[l];
    s += v;
    if (v > mx) {
      mx = v;
      mi = (int)l;
    }
  }
  y[i] = (scalar_t)(s / (float)(lo1 - lo0) + mx);
  argmax[i] = mi;
  }
}

template <typename scalar_t>
__global__ void avgmax_bwd_kernel(const scalar_t* __restrict__ dy,
                                  const int* __restrict__ argmax,
                                  scalar_t* __restrict__ dx,
                                  long L, long Lo, int k, long rows) {
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long i = ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
    if (i >= rows * L) return;
    const long row = i / L;
    const long li = i - row * L;
    const long lo = li / k;
    const long lo0 = lo * k;
    const long lo1 = min(lo0 + (long)k, L);
    const long oi = row * Lo + lo;
    const float g = (float)dy[oi];
    float v = g / (float)(lo1 - lo0);
    if (argmax[oi] == (int)li) v += g;
    dx[i] = (scalar_t)v;
  }
}

template <typename scalar_t>
__global__ void interp_fwd_kernel(const scalar_t* __restrict__ x,
                                  scalar_t* __restrict__ y,
                                  long Li, long Lo, float scale, long rows) {
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
  const long i = ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
  if (i >= rows * Lo) return;
  const long row = i / Lo;
  const long lo = i - row * Lo;
  float src = ((float)lo + 0.5f) * scale - 0.5f;
  src = fmaxf(src, 0.0f);
  long l0 = (long)src;
  l0 = min(l0, Li - 1);
  const long l1 = min(l0 + 1, Li - 1);
  const float w1 = src - (float)l0;
  const scalar_t* xr = x + row * Li;
  y[i] = (scalar_t)((1.0f - w1) * (float)xr[l0] + w1 * (float)xr[l1]);
  }
}

// Gather formulation: each dx element sums the dy window that maps onto
// it (deterministic, no atomics — the scatter version serialised on
// fp32 atomics when Lo >> Li, e.g. the 8192 -> 64 upsampling-head
// backward).
template <typename scalar_t>
__global__ void interp_bwd_kernel(const scalar_t* __restrict__ dy,
                                  scalar_t* __restrict__ dx,
                                  long Li, long Lo, float scale, long rows) {
  const long i = (long)blockIdx.x * kBlock + threadIdx.x;
  if (i >= rows * Li) return;
  const long row = i / Li;
  const long li = i - row * Li;
  // dy[lo] touches dx indices l0(lo) and l0(lo)+1 where
  // src(lo) = max((lo+0.5)*scale - 0.5, 0), l0 = min(floor(src), Li-1).
  // Conservative window of lo whose src lies in (li-1, li+1):
  const float inv = 1.0f / scale;
  long lo_lo = (long)floorf(((float)li - 1.0f + 0.5f) * inv - 0.5f) - 1;
  long lo_hi = (long)ceilf(((float)li + 1.0f + 0.5f) * inv - 0.5f) + 1;
  if (li == 0) lo_lo = 0;           // src clamp maps early lo to l0 = 0
  lo_lo = max(lo_lo, (long)0);
  lo_hi = min(lo_hi, Lo - 1);
  const scalar_t* dyr = dy + row * Lo;
  float acc = 0.0f;
  for (long lo = lo_lo; lo <= lo_hi; ++lo) {
    float src = ((float)lo + 0.5f) * scale - 0.5f;
    src = fmaxf(src, 0.0f);
    long l0 = (long)src;
    l0 = min(l0, Li - 1);
    const long l1 = min(l0 + 1, Li - 1);
    const float w1 = src - (float)l0;
    float wgt = 0.0f;
    if (l0 == li) wgt += 1.0f - w1;
    if (l1 == li && w1 != 0.0f) wgt += w1;
    acc += wgt * (float)dyr[lo];
  }
  dx[i] = (scalar_t)acc;
}

// LDS-staged variant: one block per (row, 256-wide li chunk); the dy window
// the chunk gathers from is staged once with coalesced loads, so the
// ~1/scale-wide inner loop reads LDS instead of issuing one global load per
// tap (the gather version was VMEM-issue bound, ~8x off roofline on the
// 2x head upsamples). Falls back to the gather kernel for extreme ratios
// whose window would not fit in LDS.
template <typename scalar_t>
__global__ void interp_bwd_lds_kernel(const scalar_t* __restrict__ dy,
                                      scalar_t* __restrict__ dx,
                                      long Li, long Lo, float scale,
                                      long rows) {
  extern __shared__ float dy_s[];
  const long row = blockIdx.y;
  const long li0 = (long)blockIdx.x * (kBlock * kEwTile);
  const float inv = 1.0f / scale;
  long blo = (long)floorf(((float)li0 - 0.5f) * inv - 0.5f) - 1;
  blo = max(blo, (long)0);
  long bhi = (long)ceilf(
                 ((float)(li0 + kBlock * kEwTile - 1) + 1.5f) * inv - 0.5f)
             + 1;
  bhi = min(bhi, Lo - 1);
  const int ext = (int)(bhi - blo + 1);
  const scalar_t* dyr = dy + row * Lo;
  for (int idx = threadIdx.x; idx < ext; idx += kBlock) {
    dy_s[idx] = (float)dyr[blo + idx];
  }
  __syncthreads();
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long li = li0 + t * kBlock + threadIdx.x;
    if (li >= Li) break;
    if (Lo == 2 * Li && li > 0 && li < Li - 1) {
      // exact 2x upsample (every power-of-two head stage): closed 4-tap
      // form; dy[2li]/dy[2li+1] carry 0.75, the outer neighbours 0.25
      const long b = 2 * li - blo;
      const float acc2 = 0.25f * dy_s[b - 1] + 0.75f * dy_s[b]
                         + 0.75f * dy_s[b + 1] + 0.25f * dy_s[b + 2];
      dx[row * Li + li] = (scalar_t)acc2;
      continue;
    }
    long lo_lo = (long)floorf(((float)li - 1.0f + 0.5f) * inv - 0.5f) - 1;
    long lo_hi = (long)ceilf(((float)li + 1.0f + 0.5f) * inv - 0.5f) + 1;
    if (li == 0) lo_lo = 0;
    lo_lo = max(lo_lo, blo);
    lo_hi = min(lo_hi, bhi);
    float acc = 0.0f;
    for (long lo = lo_lo; lo <= lo_hi; ++lo) {
      float src = ((float)lo + 0.5f) * scale - 0.5f;
      src = fmaxf(src, 0.0f);
      long l0 = (long)src;
      l0 = min(l0, Li - 1);
      const long l1 = min(l0 + 1, Li - 1);
      const float w1 = src - (float)l0;
      float wgt = 0.0f;
      if (l0 == li) wgt += 1.0f - w1;
      if (l1 == li && w1 != 0.0f) wgt += w1;
      acc += wgt * dy_s[lo - blo];
    }
    dx[row * Li + li] = (scalar_t)acc;
  }
}

template <typename scalar_t>
__global__ void maxpool_fwd_kernel(const scalar_t* __restrict__ x,
                                   scalar_t* __restrict__ y,
                                   int* __restrict__ argmax,
                                   long L, long Lo, int k, long rows) {
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long i = ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
    if (i >= rows * Lo) return;
    const long row = i / Lo;
    const long lo = i - row * Lo;
    const long lo0 = lo * k;
    const long lo1 = min(lo0 + (long)k, L);
    const scalar_t* xr = x + row * L;
    float mx = -INFINITY;
    int mi = (int)lo0;
    for (long l = lo0; l < lo1; ++l) {
      const float v = (float)xr[l];
      if (v > mx) {
        mx = v;
        mi = (int)l;
      }
    }
    y[i] = (scalar_t)mx;
    argmax[i] = mi;
  }
}

template <typename scalar_t>
__global__ void maxpool_bwd_kernel(const scalar_t* __restrict__ dy,
                                   const int* __restrict__ argmax,
                                   scalar_t* __restrict__ dx,
                                   long L, long Lo, int k, long rows) {
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long i = ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
    if (i >= rows * L) return;
    const long row = i / L;
    const long li = i - row * L;
    const long lo = li / k;
    const long oi = row * Lo + lo;
    dx[i] = (argmax[oi] == (int)li) ? dy[oi] : (scalar_t)0.0f;
  }
}

// global average pool to length 1 (AdaptiveAvgPool1d(1)): one wave per row
template <typename scalar_t>
__global__ void gap_fwd_kernel(const scalar_t* __restrict__ x,
                               scalar_t* __restrict__ y, long L, long rows) {
  const int lane = threadIdx.x & (sa::kWave - 1);
  const long row = (long)blockIdx.x * (blockDim.x / sa::kWave)
                   + threadIdx.x / sa::kWave;
  if (row >= rows) return;
  const scalar_t* xr = x + row * L;
  float s = 0.0f;
  for (long l = lane; l < L; l += sa::kWave) s += (float)xr[l];
  s = sa::warp_reduce_sum(s);
  if (lane == 0) y[row] = (scalar_t)(s / (float)L);
}

template <typename scalar_t>
__global__ void gap_bwd_kernel(const scalar_t* __restrict__ dy,
                               scalar_t* __restrict__ dx, long L, long rows) {
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long i = ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
    if (i >= rows * L) return;
    const long row = i / L;
    dx[i] = (scalar_t)((float)dy[row] / (float)L);
  }
}

}  // namespace

std::vector<at::Tensor> avgmax_pool_fwd(const at::Tensor& x, long k) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  const long rows = (long)x.size(0) * x.size(1);
  const long L = x.size(2);
  const long Lo = (L + k - 1) / k;
  auto y = at::empty({x.size(0), x.size(1), Lo}, x.options());
  auto idx = at::empty({x.size(0), x.size(1), Lo},
                       x.options().dtype(at::kInt));
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "avgmax_fwd", [&] {
        hipLaunchKernelGGL((avgmax_fwd_kernel<scalar_t>),
                           dim3(sa::ceil_div(rows * Lo,
                                             (long)kBlock * kEwTile)),
                           dim3(kBlock), 0, stream.stream(),
                           x.data_ptr<scalar_t>(), y.data_ptr<scalar_t>(),
                           idx.data_ptr<int>(), L, Lo, (int)k, rows);
      });
  return {y, idx};
}

at::Tensor avgmax_pool_bwd(const at::Tensor& dy, const at::Tensor& argmax,
                           long k, long in_len) {
  const long rows = (long)dy.size(0) * dy.size(1);
  const long Lo = dy.size(2);
  auto dx = at::empty({dy.size(0), dy.size(1), in_len}, dy.options());
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, dy.scalar_type(),
      "avgmax_bwd", [&] {
        hipLaunchKernelGGL((avgmax_bwd_kernel<scalar_t>),
                           dim3(sa::ceil_div(rows * in_len,
                                             (long)kBlock * kEwTile)),
                           dim3(kBlock), 0, stream.stream(),
                           dy.data_ptr<scalar_t>(), argmax.data_ptr<int>(),
                           dx.data_ptr<scalar_t>(), in_len, Lo, (int)k, rows);
      });
  return dx;
}

at::Tensor interp_linear_fwd(const at::Tensor& x, long out_len) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  const long rows = (long)x.size(0) * x.size(1);
  const long Li = x.size(2);
  auto y = at::empty({x.size(0), x.size(1), out_len}, x.options());
  const float scale = (float)Li / (float)out_len;
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "interp_fwd", [&] {
        hipLaunchKernelGGL((interp_fwd_kernel<scalar_t>),
                           dim3(sa::ceil_div(rows * out_len,
                                             (long)kBlock * kEwTile)),
                           dim3(kBlock), 0, stream.stream(),
                           x.data_ptr<scalar_t>(), y.data_ptr<scalar_t>(),
                           Li, out_len, scale, rows);
      });
  return y;
}

at::Tensor interp_linear_bwd(const at::Tensor& dy, long in_len) {
  const long rows = (long)dy.size(0) * dy.size(1);
  const long Lo = dy.size(2);
  auto dx = at::empty({dy.size(0), dy.size(1), in_len}, dy.options());
  const float scale = (float)in_len / (float)Lo;
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, dy.scalar_type(),
      "interp_bwd", [&] {
        // dy window of one 256-li block, with slack for the edge clamps
        const float inv = 1.0f / scale;
        const long ext_bound = (long)((kBlock * kEwTile + 4) * inv) + 8;
        if (ext_bound <= 12288 && rows <= 65535) {
          dim3 grid(sa::ceil_div(in_len, (long)kBlock * kEwTile), rows);
          hipLaunchKernelGGL((interp_bwd_lds_kernel<scalar_t>), grid,
                             dim3(kBlock), sizeof(float) * ext_bound,
                             stream.stream(), dy.data_ptr<scalar_t>(),
                             dx.data_ptr<scalar_t>(), in_len, Lo, scale,
                             rows);
        } else {
          hipLaunchKernelGGL((interp_bwd_kernel<scalar_t>),
                             dim3(sa::ceil_div(rows * in_len, kBlock)),
                             dim3(kBlock), 0, stream.stream(),
                             dy.data_ptr<scalar_t>(),
                             dx.data_ptr<scalar_t>(), in_len, Lo, scale,
                             rows);
        }
      });
  return dx;
}

// Nearest-neighbour x2 upsample (EQTransformer decoders,
// reference eqtransformer.py:384): forward duplicates each sample,
// backward sums each output pair — both single passes (ATen's generic
// upsample_nearest1d backward measured ~5x off the roofline here).

namespace {

template <typename scalar_t>
__global__ void up2_fwd_kernel(const scalar_t* __restrict__ x,
                               scalar_t* __restrict__ y,
                               long Li, long rows) {
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long i = ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
    if (i >= rows * Li) return;
    const long row = i / Li;
    const long li = i - row * Li;
    const scalar_t v = x[i];
    scalar_t* yr = y + row * Li * 2 + li * 2;
    yr[0] = v;
    yr[1] = v;
  }
}

template <typename scalar_t>
__global__ void up2_bwd_kernel(const scalar_t* __restrict__ dy,
                               scalar_t* __restrict__ dx,
                               long Li, long rows) {
#pragma unroll
  for (int t = 0; t < kEwTile; ++t) {
    const long i = ((long)blockIdx.x * kEwTile + t) * kBlock + threadIdx.x;
    if (i >= rows * Li) return;
    const long row = i / Li;
    const long li = i - row * Li;
    const scalar_t* dyr = dy + row * Li * 2 + li * 2;
    dx[i] = (scalar_t)((float)dyr[0] + (float)dyr[1]);
  }
}

}  // namespace

at::Tensor upsample2x_fwd(const at::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  const long rows = (long)x.size(0) * x.size(1);
  const long Li = x.size(2);
  auto y = at::empty({x.size(0), x.size(1), Li * 2}, x.options());
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "up2_fwd", [&] {
        hipLaunchKernelGGL((up2_fwd_kernel<scalar_t>),
                           dim3(sa::ceil_div(rows * Li,
                                             (long)kBlock * kEwTile)),
                           dim3(kBlock), 0, stream.stream(),
                           x.data_ptr<scalar_t>(), y.data_ptr<scalar_t>(),
                           Li, rows);
      });
  return y;
}

at::Tensor upsample2x_bwd(const at::Tensor& dy) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && dy.dim() == 3);
  const long Lo = dy.size(2);
  const long Li = Lo / 2;
  const long rows = (long)dy.size(0) * dy.size(1);
  auto dx = at::empty({dy.size(0), dy.size(1), Li}, dy.options());
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, dy.scalar_type(),
      "up2_bwd", [&] {
        hipLaunchKernelGGL((up2_bwd_kernel<scalar_t>),
                           dim3(sa::ceil_div(rows * Li,
                                             (long)kBlock * kEwTile)),
                           dim3(kBlock), 0, stream.stream(),
                           dy.data_ptr<scalar_t>(), dx.data_ptr<scalar_t>(),
                           Li, rows);
      });
  return dx;
}


std::vector<at::Tensor> max_pool1d_fwd(const at::Tensor& x, long k,
                                       bool ceil_mode) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  const long L = x.size(2);
  const long rows = x.size(0) * x.size(1);
  const long Lo = ceil_mode ? (L + k - 1) / k : L / k;
  auto y = at::empty({x.size(0), x.size(1), Lo}, x.options());
  auto argmax = at::empty({x.size(0), x.size(1), Lo},
                          x.options().dtype(at::kInt));
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "max_pool1d_fwd", [&] {
        hipLaunchKernelGGL(
            (maxpool_fwd_kernel<scalar_t>),
            dim3(sa::ceil_div(rows * Lo, (long)kBlock * kEwTile)),
            dim3(kBlock), 0, stream.stream(), x.data_ptr<scalar_t>(),
            y.data_ptr<scalar_t>(), argmax.data_ptr<int>(), L, Lo, (int)k,
            rows);
      });
  return {y, argmax};
}

at::Tensor max_pool1d_bwd(const at::Tensor& dy, const at::Tensor& argmax,
                          long k, long in_len) {
  const long rows = dy.size(0) * dy.size(1);
  const long Lo = dy.size(2);
  auto dx = at::empty({dy.size(0), dy.size(1), in_len}, dy.options());
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, dy.scalar_type(),
      "max_pool1d_bwd", [&] {
        hipLaunchKernelGGL(
            (maxpool_bwd_kernel<scalar_t>),
            dim3(sa::ceil_div(rows * in_len, (long)kBlock * kEwTile)),
            dim3(kBlock), 0, stream.stream(), dy.data_ptr<scalar_t>(),
            argmax.data_ptr<int>(), dx.data_ptr<scalar_t>(), in_len, Lo,
            (int)k, rows);
      });
  return dx;
}

at::Tensor gap_fwd(const at::Tensor& x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && x.dim() == 3);
  const long L = x.size(2);
  const long rows = x.size(0) * x.size(1);
  auto y = at::empty({x.size(0), x.size(1), 1}, x.options());
  auto stream = at::hip::getCurrentHIPStream();
  const int rpb = kBlock / sa::kWave;
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "gap_fwd", [&] {
        hipLaunchKernelGGL((gap_fwd_kernel<scalar_t>),
                           dim3(sa::ceil_div(rows, (long)rpb)), dim3(kBlock),
                           0, stream.stream(), x.data_ptr<scalar_t>(),
                           y.data_ptr<scalar_t>(), L, rows);
      });
  return y;
}

at::Tensor gap_bwd(const at::Tensor& dy, long in_len) {
  const long rows = dy.size(0) * dy.size(1);
  auto dx = at::empty({dy.size(0), dy.size(1), in_len}, dy.options());
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, dy.scalar_type(),
      "gap_bwd", [&] {
        hipLaunchKernelGGL(
            (gap_bwd_kernel<scalar_t>),
            dim3(sa::ceil_div(rows * in_len, (long)kBlock * kEwTile)),
            dim3(kBlock), 0, stream.stream(), dy.data_ptr<scalar_t>(),
            dx.data_ptr<scalar_t>(), in_len, rows);
      });
  return dx;
}
