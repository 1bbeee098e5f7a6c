// Fused pooled-KV attention forward — K9 of SURVEY.md §2.4, the SeisT
// signature op (reference models/seist.py:368-393): out = softmax(
// (q/sqrt(E))^T k) @ v^T with K/V of pooled length Lk (128 at every stage
// of the published configs).
//
// Inference path: one kernel per (n, h) pair — K and V live in LDS, each
// thread owns one query column and runs the online-softmax accumulation
// (score, rescale, accumulate) in registers, so neither the (Lq x Lk)
// attention matrix nor any intermediate ever touches HBM. Training uses
// the rocBLAS bmm + fused-softmax composite (attention dropout needs RNG
// state that autograd replays).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kMaxE = 32;
constexpr int kMaxLk = 256;

template <typename scalar_t, int E>
__global__ __launch_bounds__(kBlock)
void pooled_attn_kernel(const scalar_t* __restrict__ q,
                        const scalar_t* __restrict__ k,
                        const scalar_t* __restrict__ v,
                        scalar_t* __restrict__ out,
                        long Lq, int Lk, float scale) {
  extern __shared__ float kv_s[];  // [2][E][Lk]
  const long nh = blockIdx.y;
  const long lq = (long)blockIdx.x * kBlock + threadIdx.x;

  float* k_s = kv_s;
  float* v_s = kv_s + E * Lk;
  const scalar_t* kb = k + nh * (long)E * Lk;
  const scalar_t* vb = v + nh * (long)E * Lk;
  for (int idx = threadIdx.x; idx < E * Lk; idx += kBlock) {
    k_s[idx] = (float)kb[idx];
    v_s[idx] = (float)vb[idx];
  }
  __syncthreads();
  if (lq >= Lq) return;

  float qr[E];
  const scalar_t* qb = q + nh * (long)E * Lq + lq;
#pragma unroll
  for (int e = 0; e < E; ++e) qr[e] = (float)qb[(long)e * Lq] * scale;

  float m = -INFINITY;
  float denom = 0.0f;
  float acc[E];
#pragma unroll
  for (int e = 0; e < E; ++e) acc[e] = 0.0f;

  for (int j = 0; j < Lk; ++j) {
    float s = 0.0f;
#pragma unroll
    for (int e = 0; e < E; ++e) s += qr[e] * k_s[e * Lk + j];
    if (s > m) {
      const float c = (m == -INFINITY) ? 0.0f : __expf(m - s);
      denom *= c;
#pragma unroll
      for (int e = 0; e < E; ++e) acc[e] *= c;
      m = s;
    }
    const float p = __expf(s - m);
    denom += p;
#pragma unroll
    for (int e = 0; e < E; ++e) acc[e] += p * v_s[e * Lk + j];
  }

  scalar_t* ob = out + nh * (long)E * Lq + lq;
  const float inv = 1.0f / denom;
#pragma unroll
  for (int e = 0; e < E; ++e) {
    ob[(long)e * Lq] = (scalar_t)(acc[e] * inv);
  }
}

}  // namespace

at::Tensor pooled_attn_fwd(const at::Tensor& q, const at::Tensor& k,
                           const at::Tensor& v) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous()
              && v.is_contiguous());
  TORCH_CHECK(q.dim() == 4, "expected (N, H, E, Lq)");
  const long N = q.size(0), H = q.size(1);
  const int E = q.size(2);
  const long Lq = q.size(3);
  const int Lk = k.size(3);
  TORCH_CHECK(E <= kMaxE && Lk <= kMaxLk,
              "pooled_attn_fwd: E<=32 and Lk<=256 supported");
  auto out = at::empty_like(q);
  const float scale = 1.0f / std::sqrt((float)E);
  const size_t lds = sizeof(float) * 2 * E * Lk;
  dim3 grid(sa::ceil_div(Lq, kBlock), N * H);
  auto stream = at::hip::getCurrentHIPStream();
  TORCH_CHECK(E == 8 || E == 16 || E == 32,
              "pooled_attn_fwd: head_dim must be 8/16/32");
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, q.scalar_type(),
      "pooled_attn_fwd", [&] {
        auto launch = [&](auto e_) {
          hipLaunchKernelGGL((pooled_attn_kernel<scalar_t,
                                                 decltype(e_)::value>),
                             grid, dim3(kBlock), lds, stream.stream(),
                             q.data_ptr<scalar_t>(), k.data_ptr<scalar_t>(),
                             v.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(),
                             Lq, Lk, scale);
        };
        if (E == 8) launch(std::integral_constant<int, 8>{});
        else if (E == 16) launch(std::integral_constant<int, 16>{});
        else launch(std::integral_constant<int, 32>{});
      });
  return out;
}
