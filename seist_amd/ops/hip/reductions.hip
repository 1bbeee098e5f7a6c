// Small dedicated reductions used by the backward passes:
//  * sum_batch:   (B, ...) -> (...)   fp32, batch axis reduced
//  * channel_sum: (N, C, L) -> (C)    fp32 (bias gradients)
// Both replace generic at::sum calls that showed up as ~8 ms/step in the
// training profile (220+ launches of ATen reduce_kernel per step).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

namespace {

constexpr int kBlock = 256;

template <typename scalar_t>
__global__ void sum_batch_kernel(const scalar_t* __restrict__ in,
                                 float* __restrict__ part,  // (nsplit, M)
                                 long B, long M) {
  const long j = (long)blockIdx.x * kBlock + threadIdx.x;
  if (j >= M) return;
  const int nsplit = gridDim.y;
  const long bchunk = (B + nsplit - 1) / nsplit;
  const long b0 = (long)blockIdx.y * bchunk;
  const long b1 = min(B, b0 + bchunk);
  float s = 0.0f;
  const scalar_t* p = in + j;
  long b = b0;
  for (; b + 4 <= b1; b += 4) {
    s += (float)p[b * M] + (float)p[(b + 1) * M]
         + (float)p[(b + 2) * M] + (float)p[(b + 3) * M];
  }
  for (; b < b1; ++b) s += (float)p[b * M];
  part[(long)blockIdx.y * M + j] = s;
}

// 64 columns x 4 split-slices per block: M is often small (Co*Ci of one
// conv), so column-only parallelism leaves the chip idle — the 4 z-slices
// quadruple the block count and the per-thread serial depth drops to
// nsplit/4.
template <typename out_t>
__global__ void sum_batch_final_kernel(const float* __restrict__ part,
                                       out_t* __restrict__ out,
                                       int nsplit, long M) {
  __shared__ float red[8][32];
  const int jt = threadIdx.x & 31;
  const int zs = threadIdx.x >> 5;
  const long j = (long)blockIdx.x * 32 + jt;
  float s = 0.0f;
  if (j < M) {
    for (int i = zs; i < nsplit; i += 8) s += part[(long)i * M + j];
  }
  red[zs][jt] = s;
  __syncthreads();
  if (threadIdx.x < 32 && j < M) {
    float t = 0.0f;
#pragma unroll
    for (int z = 0; z < 8; ++z) t += red[z][jt];
    out[j] = (out_t)t;
  }
}

template <typename scalar_t>
__global__ void channel_sum_kernel(const scalar_t* __restrict__ in,
                                   float* __restrict__ part,  // (C, nsplit)
                                   int C, long N, long L) {
  __shared__ float red[kBlock / sa::kWave];
  const int c = blockIdx.x;
  const int split = blockIdx.y;
  const int nsplit = gridDim.y;
  const long nchunk = (N + nsplit - 1) / nsplit;
  const long n0 = (long)split * nchunk;
  const long n1 = min(N, n0 + nchunk);
  float s = 0.0f;
  for (long n = n0; n < n1; ++n) {
    const scalar_t* r = in + (n * C + c) * L;
    for (long l = threadIdx.x; l < L; l += kBlock) s += (float)r[l];
  }
  s = sa::block_reduce_sum(s, red);
  if (threadIdx.x == 0) part[(long)c * nsplit + split] = s;
}

// one wave per channel (a thread-per-channel serial loop is latency-bound)
template <typename out_t>
__global__ void part_sum_kernel(const float* __restrict__ part,
                                out_t* __restrict__ out, int C, int nsplit) {
  const int lane = threadIdx.x & (sa::kWave - 1);
  const int c = blockIdx.x * (blockDim.x / sa::kWave)
                + threadIdx.x / sa::kWave;
  if (c >= C) return;
  float s = 0.0f;
  for (int j = lane; j < nsplit; j += sa::kWave) {
    s += part[(long)c * nsplit + j];
  }
  s = sa::warp_reduce_sum(s);
  if (lane == 0) out[c] = (out_t)s;
}

template <typename scalar_t, typename out_t>
__global__ void sum_mid_kernel(const scalar_t* __restrict__ in,
                               out_t* __restrict__ out,
                               long A, long B, long M) {
  const long i = (long)blockIdx.x * kBlock + threadIdx.x;  // a*M + j
  if (i >= A * M) return;
  const long a = i / M;
  const long j = i - a * M;
  const scalar_t* p = in + (a * B) * M + j;
  float s = 0.0f;
  long b = 0;
  for (; b + 4 <= B; b += 4) {
    s += (float)p[b * M] + (float)p[(b + 1) * M]
         + (float)p[(b + 2) * M] + (float)p[(b + 3) * M];
  }
  for (; b < B; ++b) s += (float)p[b * M];
  out[i] = (out_t)s;
}

}  // namespace

// (A, B, M) -> (A, M) fp32, middle axis reduced — used to collapse the
// batch axis of all K tap-GEMMs of a conv weight gradient in one launch.
at::Tensor sum_mid_to(const at::Tensor& in, at::ScalarType out_dtype) {
  TORCH_CHECK(in.is_cuda() && in.is_contiguous() && in.dim() >= 3);
  const long A = in.size(0);
  const long B = in.size(1);
  const long M = in.numel() / (A * B);
  auto out = at::empty({A, M}, in.options().dtype(out_dtype));
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, in.scalar_type(),
      "sum_mid", [&] {
        using in_t = scalar_t;
        AT_DISPATCH_FLOATING_TYPES_AND2(
            at::ScalarType::BFloat16, at::ScalarType::Half, out_dtype,
            "sum_mid_out", [&] {
              using out_t = scalar_t;
              hipLaunchKernelGGL((sum_mid_kernel<in_t, out_t>),
                                 dim3(sa::ceil_div(A * M, kBlock)),
                                 dim3(kBlock), 0, stream.stream(),
                                 in.data_ptr<in_t>(),
                                 out.data_ptr<out_t>(), A, B, M);
            });
      });
  return out;
}

at::Tensor sum_mid(const at::Tensor& in) {
  return sum_mid_to(in, at::kFloat);
}

at::Tensor sum_batch_to(const at::Tensor& in, at::ScalarType out_dtype) {
  TORCH_CHECK(in.is_cuda() && in.is_contiguous() && in.dim() >= 2);
  const long B = in.size(0);
  const long M = in.numel() / B;
  auto out_sizes = in.sizes().vec();
  out_sizes.erase(out_sizes.begin());
  auto out = at::empty(out_sizes, in.options().dtype(out_dtype));
  auto stream = at::hip::getCurrentHIPStream();
  // target ~512 blocks total, nsplit bounded so the final pass stays tiny
  const long col_blocks = std::max<long>(M / kBlock, 1);
  const int nsplit = std::max(1, (int)std::min<long>(
      std::min<long>(B, 128), 1024 / col_blocks));
  auto part = at::empty({nsplit, M}, in.options().dtype(at::kFloat));
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, in.scalar_type(),
      "sum_batch", [&] {
        hipLaunchKernelGGL((sum_batch_kernel<scalar_t>),
                           dim3(sa::ceil_div(M, kBlock), nsplit),
                           dim3(kBlock), 0,
                           stream.stream(), in.data_ptr<scalar_t>(),
                           part.data_ptr<float>(), B, M);
      });
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, out_dtype,
      "sum_batch_final", [&] {
        hipLaunchKernelGGL((sum_batch_final_kernel<scalar_t>),
                           dim3(sa::ceil_div(M, 32)), dim3(kBlock), 0,
                           stream.stream(), part.data_ptr<float>(),
                           out.data_ptr<scalar_t>(), nsplit, M);
      });
  return out;
}

at::Tensor sum_batch(const at::Tensor& in) {
  return sum_batch_to(in, at::kFloat);
}

at::Tensor channel_sum_to(const at::Tensor& in, at::ScalarType out_dtype) {
  TORCH_CHECK(in.is_cuda() && in.is_contiguous() && in.dim() == 3);
  const long N = in.size(0);
  const int C = in.size(1);
  const long L = in.size(2);
  const int nsplit = std::max(1, std::min<int>(
      (int)N, 2048 / std::max(C, 1)));
  auto part = at::empty({C, nsplit}, in.options().dtype(at::kFloat));
  auto out = at::empty({C}, in.options().dtype(out_dtype));
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, in.scalar_type(),
      "channel_sum", [&] {
        hipLaunchKernelGGL((channel_sum_kernel<scalar_t>), dim3(C, nsplit),
                           dim3(kBlock), 0, stream.stream(),
                           in.data_ptr<scalar_t>(), part.data_ptr<float>(),
                           C, N, L);
      });
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, out_dtype,
      "part_sum", [&] {
        hipLaunchKernelGGL((part_sum_kernel<scalar_t>),
                           dim3(sa::ceil_div(C, 4)), dim3(256), 0,
                           stream.stream(), part.data_ptr<float>(),
                           out.data_ptr<scalar_t>(), C, nsplit);
      });
  return out;
}

at::Tensor channel_sum(const at::Tensor& in) {
  return channel_sum_to(in, at::kFloat);
}
