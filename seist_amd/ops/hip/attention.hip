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

// ---------------------------------------------------------------------------
// Training path: fused forward with saved softmax stats and a bit-packed
// dropout mask (32x smaller than the (Lq x Lk) probability tensor the
// bmm+softmax composite materializes), plus a flash-style backward split
// into a per-query kernel (dQ + the softmax correction term) and a per-key
// kernel (dK, dV). The mask is saved, not replayed from RNG state, so the
// backward is exactly reproducible and directly testable.
//
// RNG: a device-resident seed word bumped by a 1-thread kernel after every
// forward — under hipGraph capture the bump replays too, so each graph
// replay draws fresh masks (a host-side seed would freeze in the graph).
// ---------------------------------------------------------------------------

namespace {

constexpr int kLqChunk = 32;   // per-key backward: staged query chunk

__device__ __forceinline__ float rng_uniform(unsigned long long seed,
                                             unsigned long long idx) {
  // splitmix64 counter hash — i.i.d. enough for dropout masks
  unsigned long long z = seed + idx * 0x9E3779B97F4A7C15ull;
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ull;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBull;
  z = z ^ (z >> 31);
  return (float)(z >> 40) * (1.0f / 16777216.0f);
}

__global__ void bump_seed_kernel(unsigned long long* s) {
  *s += 0x9E3779B97F4A7C15ull;
}

template <typename scalar_t, int E, bool DROP>
__global__ __launch_bounds__(kBlock)
void pooled_attn_tfwd_kernel(const scalar_t* __restrict__ q,
                             const scalar_t* __restrict__ k,
                             const scalar_t* __restrict__ v,
                             scalar_t* __restrict__ out,
                             float* __restrict__ stats,      // (NH, Lq, 2)
                             unsigned* __restrict__ mask,    // (NH, Lq, W)
                             const unsigned long long* seed_ptr,
                             long Lq, int Lk, int W, float scale,
                             float p, float inv_keep) {
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

  const unsigned long long seed = DROP ? *seed_ptr : 0ull;
  const unsigned long long ridx0 = DROP ? (nh * Lq + lq) * (unsigned long long)Lk : 0ull;

  float qr[E];
  const scalar_t* qb = q + nh * (long)E * Lq + lq;
#pragma unroll
  for (int e = 0; e < E; ++e) qr[e] = (float)qb[(long)e * Lq] * scale;

  float m = -INFINITY;
  float denom = 0.0f;
  float acc[E];
#pragma unroll
  for (int e = 0; e < E; ++e) acc[e] = 0.0f;
  unsigned word = 0u;

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
    const float pexp = __expf(s - m);
    denom += pexp;
    float contrib = pexp;
    if (DROP) {
      const bool keep = rng_uniform(seed, ridx0 + j) >= p;
      if (keep) word |= 1u << (j & 31);
      contrib = keep ? pexp * inv_keep : 0.0f;
      if ((j & 31) == 31) {
        mask[(nh * Lq + lq) * W + (j >> 5)] = word;
        word = 0u;
      }
    }
#pragma unroll
    for (int e = 0; e < E; ++e) acc[e] += contrib * v_s[e * Lk + j];
  }
  if (DROP && (Lk & 31)) {
    mask[(nh * Lq + lq) * W + ((Lk - 1) >> 5)] = word;
  }

  stats[(nh * Lq + lq) * 2 + 0] = m;
  stats[(nh * Lq + lq) * 2 + 1] = denom;
  scalar_t* ob = out + nh * (long)E * Lq + lq;
  const float inv = 1.0f / denom;
#pragma unroll
  for (int e = 0; e < E; ++e) {
    ob[(long)e * Lq] = (scalar_t)(acc[e] * inv);
  }
}

// per-query: delta[lq] = sum_e dOut*Out (== the softmax correction term,
// dropout included), dQ[e,lq] = scale * sum_j dS[lq,j] * K[e,j]
template <typename scalar_t, int E, bool DROP>
__global__ __launch_bounds__(kBlock)
void pooled_attn_bwd_q_kernel(const scalar_t* __restrict__ q,
                              const scalar_t* __restrict__ k,
                              const scalar_t* __restrict__ v,
                              const scalar_t* __restrict__ out,
                              const scalar_t* __restrict__ dout,
                              const float* __restrict__ stats,
                              const unsigned* __restrict__ mask,
                              scalar_t* __restrict__ dq,
                              float* __restrict__ delta_out,  // (NH, Lq)
                              long Lq, int Lk, int W, float scale,
                              float inv_keep) {
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

  const long row = nh * Lq + lq;
  float qr[E], dor[E];
  const scalar_t* qb = q + nh * (long)E * Lq + lq;
  const scalar_t* ob = out + nh * (long)E * Lq + lq;
  const scalar_t* db = dout + nh * (long)E * Lq + lq;
  float delta = 0.0f;
#pragma unroll
  for (int e = 0; e < E; ++e) {
    qr[e] = (float)qb[(long)e * Lq] * scale;
    dor[e] = (float)db[(long)e * Lq];
    delta += dor[e] * (float)ob[(long)e * Lq];
  }
  delta_out[row] = delta;

  const float m = stats[row * 2 + 0];
  const float invd = 1.0f / stats[row * 2 + 1];
  float dq_acc[E];
#pragma unroll
  for (int e = 0; e < E; ++e) dq_acc[e] = 0.0f;

  unsigned word = 0u;
  for (int j = 0; j < Lk; ++j) {
    if (DROP && (j & 31) == 0) word = mask[row * W + (j >> 5)];
    float s = 0.0f;
#pragma unroll
    for (int e = 0; e < E; ++e) s += qr[e] * k_s[e * Lk + j];
    const float P = __expf(s - m) * invd;
    float dP = 0.0f;
#pragma unroll
    for (int e = 0; e < E; ++e) dP += dor[e] * v_s[e * Lk + j];
    if (DROP) dP = (word >> (j & 31)) & 1u ? dP * inv_keep : 0.0f;
    const float dS = P * (dP - delta);
#pragma unroll
    for (int e = 0; e < E; ++e) dq_acc[e] += dS * k_s[e * Lk + j];
  }
  scalar_t* dqb = dq + nh * (long)E * Lq + lq;
#pragma unroll
  for (int e = 0; e < E; ++e) {
    dqb[(long)e * Lq] = (scalar_t)(dq_acc[e] * scale);
  }
}

// per-key: one block per (n, h); thread j owns key column j and streams the
// queries through 64-wide LDS chunks. When Lk <= 128 the block splits into
// TWO query stripes (threads 128..255 process the chunk's second half with
// their own accumulators, merged through LDS at the end) — otherwise half
// the block would idle through the whole Lq loop.
//   dV[e,j] = sum_lq Pdrop[lq,j] * dOut[e,lq]
//   dK[e,j] = scale * sum_lq dS[lq,j] * Q[e,lq]
template <typename scalar_t, int E, bool DROP, int NS>
__global__ __launch_bounds__(kBlock)
void pooled_attn_bwd_kv_kernel(const scalar_t* __restrict__ q,
                               const scalar_t* __restrict__ k,
                               const scalar_t* __restrict__ v,
                               const scalar_t* __restrict__ dout,
                               const float* __restrict__ stats,
                               const unsigned* __restrict__ mask,
                               const float* __restrict__ delta,
                               scalar_t* __restrict__ dk,
                               scalar_t* __restrict__ dv,
                               long Lq, int Lk, int W, float scale,
                               float inv_keep) {
  constexpr int kChunk = 2 * kLqChunk;  // staged queries per iteration
  __shared__ float q_s[E * kChunk];
  __shared__ float do_s[E * kChunk];
  __shared__ float md_s[kChunk * 2];
  __shared__ float delta_s[kChunk];
  __shared__ unsigned mask_s[kChunk * 8];
  __shared__ float red[(NS > 1) ? 2 * E * 128 : 1];

  const long nh = blockIdx.x;
  const int j = (NS > 1) ? (threadIdx.x & 127) : threadIdx.x;
  const int stripe = (NS > 1) ? (threadIdx.x >> 7) : 0;
  const scalar_t* qb = q + nh * (long)E * Lq;
  const scalar_t* db = dout + nh * (long)E * Lq;

  float kr[E], vr[E], dk_acc[E], dv_acc[E];
  if (j < Lk) {
#pragma unroll
    for (int e = 0; e < E; ++e) {
      kr[e] = (float)k[(nh * (long)E + e) * Lk + j];
      vr[e] = (float)v[(nh * (long)E + e) * Lk + j];
      dk_acc[e] = 0.0f;
      dv_acc[e] = 0.0f;
    }
  }

  for (long lq0 = 0; lq0 < Lq; lq0 += kChunk) {
    const int cn = (int)min((long)kChunk, Lq - lq0);
    __syncthreads();
    for (int idx = threadIdx.x; idx < E * kChunk; idx += kBlock) {
      const int e = idx / kChunk;
      const int i = idx - e * kChunk;
      const float qv = (i < cn) ? (float)qb[(long)e * Lq + lq0 + i] : 0.0f;
      const float dv_ = (i < cn) ? (float)db[(long)e * Lq + lq0 + i] : 0.0f;
      q_s[e * kChunk + i] = qv;
      do_s[e * kChunk + i] = dv_;
    }
    for (int idx = threadIdx.x; idx < cn; idx += kBlock) {
      const long row = nh * Lq + lq0 + idx;
      md_s[idx * 2 + 0] = stats[row * 2 + 0];
      md_s[idx * 2 + 1] = stats[row * 2 + 1];
      delta_s[idx] = delta[row];
      if (DROP) {
        for (int wq = 0; wq < W; ++wq) {
          mask_s[idx * 8 + wq] = mask[row * W + wq];
        }
      }
    }
    __syncthreads();
    if (j >= Lk) continue;
    const int i0 = stripe * (cn > kLqChunk || NS == 1 ? kLqChunk : cn);
    const int i1 = (NS > 1 && stripe == 0) ? min(cn, kLqChunk) : cn;
    for (int i = i0; i < i1; ++i) {
      float s = 0.0f;
#pragma unroll
      for (int e = 0; e < E; ++e) s += q_s[e * kChunk + i] * kr[e];
      const float P = __expf(s * scale - md_s[i * 2 + 0]) / md_s[i * 2 + 1];
      float dP = 0.0f;
#pragma unroll
      for (int e = 0; e < E; ++e) dP += do_s[e * kChunk + i] * vr[e];
      float Pd = P;
      if (DROP) {
        const bool keep = (mask_s[i * 8 + (j >> 5)] >> (j & 31)) & 1u;
        Pd = keep ? P * inv_keep : 0.0f;
        dP = keep ? dP * inv_keep : 0.0f;
      }
      const float dS = P * (dP - delta_s[i]);
#pragma unroll
      for (int e = 0; e < E; ++e) {
        dv_acc[e] += Pd * do_s[e * kChunk + i];
        dk_acc[e] += dS * q_s[e * kChunk + i];
      }
    }
  }

  if (NS > 1) {
    // stripe 1 parks its accumulators in LDS; stripe 0 merges
    __syncthreads();
    if (stripe == 1 && j < Lk) {
#pragma unroll
      for (int e = 0; e < E; ++e) {
        red[e * 128 + j] = dk_acc[e];
        red[(E + e) * 128 + j] = dv_acc[e];
      }
    }
    __syncthreads();
    if (stripe == 1) return;
    if (j < Lk) {
#pragma unroll
      for (int e = 0; e < E; ++e) {
        dk_acc[e] += red[e * 128 + j];
        dv_acc[e] += red[(E + e) * 128 + j];
      }
    }
  }
  if (j < Lk) {
#pragma unroll
    for (int e = 0; e < E; ++e) {
      dk[(nh * (long)E + e) * Lk + j] = (scalar_t)(dk_acc[e] * scale);
      dv[(nh * (long)E + e) * Lk + j] = (scalar_t)dv_acc[e];
    }
  }
}

unsigned long long* attn_seed_state(const at::Tensor& like) {
  static std::array<at::Tensor, 16> st;
  const int dev = like.get_device();
  if (!st[dev].defined()) {
    const long init = at::randint(std::numeric_limits<int64_t>::max(), {1},
                                  at::TensorOptions().dtype(at::kLong))
                          .item<long>();
    st[dev] = at::full({1}, init, like.options().dtype(at::kLong));
  }
  return (unsigned long long*)st[dev].data_ptr<long>();
}

}  // namespace

std::vector<at::Tensor> pooled_attn_train_fwd(const at::Tensor& q,
                                              const at::Tensor& k,
                                              const at::Tensor& v,
                                              double p) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous()
              && v.is_contiguous());
  TORCH_CHECK(q.dim() == 4, "expected (N, H, E, Lq)");
  const long N = q.size(0), H = q.size(1);
  const int E = q.size(2);
  const long Lq = q.size(3);
  const int Lk = k.size(3);
  TORCH_CHECK(E * Lk <= 4096 && Lk <= kMaxLk,
              "pooled_attn_train_fwd: E*Lk <= 4096 supported");
  TORCH_CHECK(E == 8 || E == 16 || E == 32, "head_dim must be 8/16/32");
  const bool drop = p > 0.0;
  const int W = (Lk + 31) / 32;
  auto out = at::empty_like(q);
  auto stats = at::empty({N, H, Lq, 2}, q.options().dtype(at::kFloat));
  auto mask = drop ? at::empty({N, H, Lq, (long)W},
                               q.options().dtype(at::kInt))
                   : at::empty({0}, q.options().dtype(at::kInt));
  const float scale = 1.0f / std::sqrt((float)E);
  const float inv_keep = drop ? (float)(1.0 / (1.0 - p)) : 1.0f;
  const size_t lds = sizeof(float) * 2 * E * Lk;
  dim3 grid(sa::ceil_div(Lq, kBlock), N * H);
  auto stream = at::hip::getCurrentHIPStream();
  unsigned long long* seed = attn_seed_state(q);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, q.scalar_type(),
      "pooled_attn_tfwd", [&] {
        auto launch = [&](auto e_, auto d_) {
          hipLaunchKernelGGL(
              (pooled_attn_tfwd_kernel<scalar_t, decltype(e_)::value,
                                       decltype(d_)::value>),
              grid, dim3(kBlock), lds, stream.stream(),
              q.data_ptr<scalar_t>(), k.data_ptr<scalar_t>(),
              v.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(),
              stats.data_ptr<float>(),
              drop ? (unsigned*)mask.data_ptr<int>() : nullptr, seed, Lq, Lk,
              W, scale, (float)p, inv_keep);
        };
        auto launch_e = [&](auto e_) {
          if (drop) launch(e_, std::true_type{});
          else launch(e_, std::false_type{});
        };
        if (E == 8) launch_e(std::integral_constant<int, 8>{});
        else if (E == 16) launch_e(std::integral_constant<int, 16>{});
        else launch_e(std::integral_constant<int, 32>{});
      });
  if (drop) {
    hipLaunchKernelGGL(bump_seed_kernel, dim3(1), dim3(1), 0,
                       stream.stream(), seed);
  }
  return {out, stats, mask};
}

std::vector<at::Tensor> pooled_attn_bwd(const at::Tensor& q,
                                        const at::Tensor& k,
                                        const at::Tensor& v,
                                        const at::Tensor& out,
                                        const at::Tensor& dout,
                                        const at::Tensor& stats,
                                        const at::Tensor& mask, double p) {
  const long N = q.size(0), H = q.size(1);
  const int E = q.size(2);
  const long Lq = q.size(3);
  const int Lk = k.size(3);
  const bool drop = p > 0.0;
  const int W = (Lk + 31) / 32;
  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  auto delta = at::empty({N * H, Lq}, q.options().dtype(at::kFloat));
  const float scale = 1.0f / std::sqrt((float)E);
  const float inv_keep = drop ? (float)(1.0 / (1.0 - p)) : 1.0f;
  const size_t lds = sizeof(float) * 2 * E * Lk;
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, q.scalar_type(),
      "pooled_attn_bwd", [&] {
        auto launch = [&](auto e_, auto d_) {
          dim3 gq(sa::ceil_div(Lq, kBlock), N * H);
          hipLaunchKernelGGL(
              (pooled_attn_bwd_q_kernel<scalar_t, decltype(e_)::value,
                                        decltype(d_)::value>),
              gq, dim3(kBlock), lds, stream.stream(),
              q.data_ptr<scalar_t>(), k.data_ptr<scalar_t>(),
              v.data_ptr<scalar_t>(), out.data_ptr<scalar_t>(),
              dout.data_ptr<scalar_t>(), stats.data_ptr<float>(),
              drop ? (const unsigned*)mask.data_ptr<int>() : nullptr,
              dq.data_ptr<scalar_t>(), delta.data_ptr<float>(), Lq, Lk, W,
              scale, inv_keep);
          auto launch_kv = [&](auto ns_) {
            hipLaunchKernelGGL(
                (pooled_attn_bwd_kv_kernel<scalar_t, decltype(e_)::value,
                                           decltype(d_)::value,
                                           decltype(ns_)::value>),
                dim3(N * H), dim3(kBlock), 0, stream.stream(),
                q.data_ptr<scalar_t>(), k.data_ptr<scalar_t>(),
                v.data_ptr<scalar_t>(), dout.data_ptr<scalar_t>(),
                stats.data_ptr<float>(),
                drop ? (const unsigned*)mask.data_ptr<int>() : nullptr,
                delta.data_ptr<float>(), dk.data_ptr<scalar_t>(),
                dv.data_ptr<scalar_t>(), Lq, Lk, W, scale, inv_keep);
          };
          if (Lk <= 128) launch_kv(std::integral_constant<int, 2>{});
          else launch_kv(std::integral_constant<int, 1>{});
        };
        auto launch_e = [&](auto e_) {
          if (drop) launch(e_, std::true_type{});
          else launch(e_, std::false_type{});
        };
        if (E == 8) launch_e(std::integral_constant<int, 8>{});
        else if (E == 16) launch_e(std::integral_constant<int, 16>{});
        else launch_e(std::integral_constant<int, 32>{});
      });
  return {dq, dk, dv};
}
