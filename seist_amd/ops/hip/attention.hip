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

}  // namespace

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

// public bump for other dropout-consuming kernels (rowscale.hip)
void bump_attn_seed(const at::Tensor& ref) {
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(bump_seed_kernel, dim3(1), dim3(1), 0, stream.stream(),
                     attn_seed_state(ref));
}

namespace {

}  // namespace


// ---------------------------------------------------------------------------
// MFMA training forward: the per-thread VALU loops of
// pooled_attn_tfwd_kernel measured ~25x off roofline; here QK^T and PV run
// on v_mfma_f32_16x16x32_bf16 wave tiles. One block = 4 waves, each owning
// a 16-query tile of one (n, h); K/V staged once per block. Saves the SAME
// (m, denom) stats and bit-packed dropout mask as the VALU kernel, so the
// flash-style backward kernels are unchanged.
// Envelope: bf16, E in {8,16,32}, Lk % 32 == 0, Lk <= 256, Lq % 16 == 0.
// ---------------------------------------------------------------------------

namespace {

typedef __bf16 sa_bf16_a;
typedef sa_bf16_a bf16x8a __attribute__((ext_vector_type(8)));
typedef float f32x4a __attribute__((ext_vector_type(4)));

constexpr int kAT = 16;        // queries per wave tile
constexpr int kAW = 4;         // waves per block
constexpr int kMaxNF = 16;     // Lk/16 <= 16

// NF = Lk/16 as a template constant: the accumulator array is indexed in
// fully-unrolled loops and stays in registers (a runtime bound would
// spill it to scratch)
template <int NF, bool DROP>
__global__ __launch_bounds__(256)
void pooled_attn_tfwd_mfma_kernel(const sa_bf16_a* __restrict__ q,
                                  const sa_bf16_a* __restrict__ k,
                                  const sa_bf16_a* __restrict__ v,
                                  sa_bf16_a* __restrict__ out,
                                  float* __restrict__ stats,
                                  unsigned* __restrict__ mask,
                                  const unsigned long long* seed_ptr,
                                  long Lq, int E,
                                  float scale, float p, float inv_keep) {
  constexpr int Lk = NF * 16;
  constexpr int W = (NF + 1) / 2;
  extern __shared__ sa_bf16_a smem[];
  constexpr int Lkp = Lk + 8;
  sa_bf16_a* k_s = smem;                       // [32][Lkp]
  sa_bf16_a* v_s = k_s + 32 * Lkp;             // [32][Lkp]
  sa_bf16_a* q_s = v_s + 32 * Lkp;             // [kAW][16][32]
  sa_bf16_a* p_s = q_s + kAW * 16 * 32;        // [kAW][16][Lkp]

  const long nh = blockIdx.y;
  const long lq0 = (long)blockIdx.x * (kAW * kAT);
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 15;
  const int grp = lane >> 4;           // 0..3
  const int kbase = grp * 8;

  // ---- stage K/V as [e][key] (zero-padded rows e >= E) ----
  for (int idx = tid; idx < 32 * (Lkp / 8); idx += 256) {
    const int e = idx / (Lkp / 8);
    const int c8 = (idx - e * (Lkp / 8)) * 8;
    bf16x8a kv = {}, vv = {};
    if (e < E && c8 + 8 <= Lk) {
      kv = *(const bf16x8a*)(k + (nh * E + e) * Lk + c8);
      vv = *(const bf16x8a*)(v + (nh * E + e) * Lk + c8);
    } else if (e < E) {
      for (int j = 0; j < 8; ++j) {
        if (c8 + j < Lk) {
          kv[j] = k[(nh * E + e) * Lk + c8 + j];
          vv[j] = v[(nh * E + e) * Lk + c8 + j];
        }
      }
    }
    *(bf16x8a*)(k_s + e * Lkp + c8) = kv;
    *(bf16x8a*)(v_s + e * Lkp + c8) = vv;
  }
  // ---- stage this block's 64 q rows transposed to [lq][e]
  // (e-major loop: consecutive threads read consecutive lq — coalesced;
  // the transposed LDS writes are cheap) ----
  for (int idx = tid; idx < 32 * 64; idx += 256) {
    const int e = idx >> 6;
    const int r = idx & 63;
    const long lq = lq0 + r;
    q_s[r * 32 + e] = (e < E && lq < Lq) ? q[(nh * E + e) * Lq + lq]
                                         : (sa_bf16_a)0.0f;
  }
  __syncthreads();

  // ---- scores: one MFMA per 16-key fragment ----
  const bf16x8a a_q =
      *(const bf16x8a*)(q_s + (wid * 16 + col) * 32 + kbase);
  f32x4a acc[NF];
#pragma unroll
  for (int f = 0; f < NF; ++f) {
    bf16x8a b;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      b[j] = k_s[(kbase + j) * Lkp + f * 16 + col];
    }
    acc[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
        a_q, b, (f32x4a){0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
  }

  // ---- softmax rows (row = grp*4 + r local query), dropout, P to LDS ----
  const unsigned long long seed = DROP ? *seed_ptr : 0ull;
#pragma unroll 4
  for (int r = 0; r < 4; ++r) {
    const int row = grp * 4 + r;
    const long lq = lq0 + wid * kAT + row;
    float m = -INFINITY;
#pragma unroll
    for (int f = 0; f < NF; ++f) m = fmaxf(m, acc[f][r] * scale);
#pragma unroll
    for (int b = 1; b < 16; b <<= 1) {
      m = fmaxf(m, __shfl_xor(m, b, 64));
    }
    float den = 0.0f;
#pragma unroll
    for (int f = 0; f < NF; ++f) {
      const float u = __expf(acc[f][r] * scale - m);
      acc[f][r] = u;    // reuse accumulator storage for u
      den += u;
    }
#pragma unroll
    for (int b = 1; b < 16; b <<= 1) {
      den += __shfl_xor(den, b, 64);
    }
    const float invd = 1.0f / den;
    unsigned keepbits = 0xffffffffu;   // bit f = keep(this lane's key in f)
    if (DROP) {
      keepbits = 0u;
      const unsigned long long ridx0 = (nh * Lq + lq) * (unsigned long long)Lk;
#pragma unroll
      for (int f = 0; f < NF; ++f) {
        const bool keep = rng_uniform(seed, ridx0 + f * 16 + col) >= p;
        if (keep) keepbits |= 1u << f;
      }
    }
#pragma unroll
    for (int f = 0; f < NF; ++f) {
      float pv = acc[f][r] * invd;
      if (DROP) pv = (keepbits >> f) & 1u ? pv * inv_keep : 0.0f;
      p_s[(wid * 16 + row) * Lkp + f * 16 + col] = (sa_bf16_a)pv;
    }
    if (col == 0 && lq < Lq) {
      stats[(nh * Lq + lq) * 2 + 0] = m;
      stats[(nh * Lq + lq) * 2 + 1] = den;
    }
    if (DROP) {
      // assemble 32-bit mask words from the 16-lane groups via ballots
#pragma unroll
      for (int w = 0; w < W; ++w) {
        const unsigned long long b0 =
            __ballot((keepbits >> (2 * w)) & 1u);
        const unsigned long long b1 = (2 * w + 1 < NF)
            ? __ballot((keepbits >> (2 * w + 1)) & 1u) : 0ull;
        if (col == 0 && lq < Lq) {
          const unsigned lo = (unsigned)((b0 >> (16 * grp)) & 0xffffull);
          const unsigned hi = (unsigned)((b1 >> (16 * grp)) & 0xffffull);
          mask[(nh * Lq + lq) * W + w] = lo | (hi << 16);
        }
      }
    }
  }
  __syncthreads();

  // ---- PV: out tile (16 lq x E) ----
  const int NE = (E + 15) / 16;
  f32x4a acc_o[2] = {{0.f, 0.f, 0.f, 0.f}, {0.f, 0.f, 0.f, 0.f}};
  for (int kk = 0; kk < Lk; kk += 32) {
    const bf16x8a a_p =
        *(const bf16x8a*)(p_s + (wid * 16 + col) * Lkp + kk + kbase);
    for (int ne = 0; ne < NE; ++ne) {
      bf16x8a b =
          *(const bf16x8a*)(v_s + (ne * 16 + col) * Lkp + kk + kbase);
      acc_o[ne] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_p, b, acc_o[ne], 0, 0, 0);
    }
  }
  for (int ne = 0; ne < NE; ++ne) {
    const int e = ne * 16 + col;
    if (e >= E) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long lq = lq0 + wid * kAT + grp * 4 + r;
      if (lq < Lq) {
        out[(nh * E + e) * Lq + lq] = (sa_bf16_a)acc_o[ne][r];
      }
    }
  }
}

}  // namespace

// MFMA-path host dispatch; returns false outside the envelope
bool pooled_attn_tfwd_mfma(const at::Tensor& q, const at::Tensor& k,
                           const at::Tensor& v, at::Tensor& out,
                           at::Tensor& stats, at::Tensor& mask, bool drop,
                           unsigned long long* seed, long NH, int E,
                           long Lq, int Lk, float scale, float p,
                           float inv_keep) {
  if (q.scalar_type() != at::kBFloat16) return false;
  if (Lk != 64 && Lk != 128 && Lk != 256) return false;
  if (Lq % 16 != 0) return false;
  const int Lkp = Lk + 8;
  const size_t lds = sizeof(sa_bf16_a)
      * (2 * 32 * Lkp + kAW * 16 * 32 + kAW * 16 * Lkp);
  dim3 grid(sa::ceil_div(Lq, (long)(kAW * kAT)), NH);
  auto stream = at::hip::getCurrentHIPStream();
  auto launch = [&](auto nf_, auto d_) {
    hipLaunchKernelGGL(
        (pooled_attn_tfwd_mfma_kernel<decltype(nf_)::value,
                                      decltype(d_)::value>),
        grid, dim3(256), lds, stream.stream(),
        (const sa_bf16_a*)q.data_ptr(), (const sa_bf16_a*)k.data_ptr(),
        (const sa_bf16_a*)v.data_ptr(), (sa_bf16_a*)out.data_ptr(),
        stats.data_ptr<float>(),
        drop ? (unsigned*)mask.data_ptr<int>() : nullptr, seed, Lq, E,
        scale, p, inv_keep);
  };
  auto launch_nf = [&](auto nf_) {
    if (drop) launch(nf_, std::true_type{});
    else launch(nf_, std::false_type{});
  };
  if (Lk == 64) launch_nf(std::integral_constant<int, 4>{});
  else if (Lk == 128) launch_nf(std::integral_constant<int, 8>{});
  else launch_nf(std::integral_constant<int, 16>{});
  return true;
}

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
  if (pooled_attn_tfwd_mfma(q, k, v, out, stats, mask, drop, seed, N * H,
                            E, Lq, Lk, scale, (float)p, inv_keep)) {
    if (drop) {
      hipLaunchKernelGGL(bump_seed_kernel, dim3(1), dim3(1), 0,
                         stream.stream(), seed);
    }
    return {out, stats, mask};
  }
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


// ---------------------------------------------------------------------------
// Single-pass MFMA backward: one block per (n, h) sweeps Lq in 64-query
// chunks, recomputing P from the saved (m, denom) stats (no reductions),
// emitting dQ per chunk and accumulating the dK/dV wave tiles across the
// sweep — replaces the per-query and per-key VALU kernels in one launch.
// Envelope: bf16, E in {8,16,32}, Lk in {64,128}, Lq % 16 == 0.
// ---------------------------------------------------------------------------

namespace {

// delta[nh, lq] = sum_e dout * out — lane-per-lq, coalesced per e
__global__ void attn_delta_kernel(const sa_bf16_a* __restrict__ out,
                                  const sa_bf16_a* __restrict__ dout,
                                  float* __restrict__ delta,
                                  long NHLq, long Lq, int E) {
  const long i = (long)blockIdx.x * 256 + threadIdx.x;
  if (i >= NHLq) return;
  const long nh = i / Lq;
  const long lq = i - nh * Lq;
  float s = 0.0f;
  for (int e = 0; e < E; ++e) {
    const long o = (nh * E + e) * Lq + lq;
    s += (float)dout[o] * (float)out[o];
  }
  delta[i] = s;
}

template <int NF, bool DROP>
__global__ __launch_bounds__(256)
void pooled_attn_bwd_mfma_kernel(const sa_bf16_a* __restrict__ q,
                                 const sa_bf16_a* __restrict__ k,
                                 const sa_bf16_a* __restrict__ v,
                                 const sa_bf16_a* __restrict__ out,
                                 const sa_bf16_a* __restrict__ dout,
                                 const float* __restrict__ delta,
                                 const float* __restrict__ stats,
                                 const unsigned* __restrict__ mask,
                                 sa_bf16_a* __restrict__ dq,
                                 sa_bf16_a* __restrict__ dk,
                                 sa_bf16_a* __restrict__ dv,
                                 long Lq, int E, float scale,
                                 float inv_keep) {
  constexpr int Lk = NF * 16;
  constexpr int W = (NF + 1) / 2;
  constexpr int Lkp = Lk + 8;
  extern __shared__ sa_bf16_a smem[];
  sa_bf16_a* k_s = smem;                       // [32][Lkp]
  sa_bf16_a* v_s = k_s + 32 * Lkp;             // [32][Lkp]
  sa_bf16_a* q_s = v_s + 32 * Lkp;             // [64][32]  (chunk, [lq][e])
  sa_bf16_a* dO_s = q_s + 64 * 32;             // [64][32]
  sa_bf16_a* p_s = dO_s + 64 * 32;             // [64][Lkp]
  sa_bf16_a* ds_s = p_s + 64 * Lkp;            // [64][Lkp]
  float* md_s = (float*)(ds_s + 64 * Lkp);     // [64][2] m, invden
  float* delta_s = md_s + 64 * 2;              // [64]
  unsigned* mk_s = (unsigned*)(delta_s + 64);  // [64][W]

  const long nh = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int col = lane & 15;
  const int grp = lane >> 4;
  const int kbase = grp * 8;
  const int NE = (E + 15) / 16;

  // ---- stage K/V [e][key], zero-padded ----
  for (int idx = tid; idx < 32 * (Lkp / 8); idx += 256) {
    const int e = idx / (Lkp / 8);
    const int c8 = (idx - e * (Lkp / 8)) * 8;
    bf16x8a kv = {}, vv = {};
    if (e < E && c8 + 8 <= Lk) {
      kv = *(const bf16x8a*)(k + (nh * E + e) * Lk + c8);
      vv = *(const bf16x8a*)(v + (nh * E + e) * Lk + c8);
    } else if (e < E) {
      for (int j = 0; j < 8; ++j) {
        if (c8 + j < Lk) {
          kv[j] = k[(nh * E + e) * Lk + c8 + j];
          vv[j] = v[(nh * E + e) * Lk + c8 + j];
        }
      }
    }
    *(bf16x8a*)(k_s + e * Lkp + c8) = kv;
    *(bf16x8a*)(v_s + e * Lkp + c8) = vv;
  }

  // dK/dV accumulators: tiles t = wid + 4*i over (f = t % NF, ne = t / NF)
  f32x4a acc_dv[4] = {{0,0,0,0},{0,0,0,0},{0,0,0,0},{0,0,0,0}};
  f32x4a acc_dk[4] = {{0,0,0,0},{0,0,0,0},{0,0,0,0},{0,0,0,0}};
  const int NT = NF * NE;

  for (long lq0 = 0; lq0 < Lq; lq0 += 64) {
    __syncthreads();
    // ---- stage chunk: q/dOut transposed [lq][e] with e-major loops
    // (coalesced lq reads); delta comes precomputed ----
    for (int idx = tid; idx < 32 * 64; idx += 256) {
      const int e = idx >> 6;
      const int r = idx & 63;
      const long lq = lq0 + r;
      sa_bf16_a qv = (sa_bf16_a)0.0f, dov = (sa_bf16_a)0.0f;
      if (e < E && lq < Lq) {
        qv = q[(nh * E + e) * Lq + lq];
        dov = dout[(nh * E + e) * Lq + lq];
      }
      q_s[r * 32 + e] = qv;
      dO_s[r * 32 + e] = dov;
    }
    if (tid < 64) {
      const long lq = lq0 + tid;
      float dlt = 0.0f, m = 0.0f, invd = 1.0f;
      if (lq < Lq) {
        dlt = delta[nh * Lq + lq];
        m = stats[(nh * Lq + lq) * 2 + 0];
        invd = 1.0f / stats[(nh * Lq + lq) * 2 + 1];
      }
      delta_s[tid] = dlt;
      md_s[tid * 2 + 0] = m;
      md_s[tid * 2 + 1] = invd;
    }
    if (DROP) {
      for (int idx = tid; idx < 64 * W; idx += 256) {
        const long lq = lq0 + idx / W;
        mk_s[idx] = (lq < Lq) ? mask[(nh * Lq + lq) * W + (idx % W)] : 0u;
      }
    }
    __syncthreads();

    // ---- scores + P~ and dP + dS for this wave's 16 lq rows ----
    const bf16x8a a_q =
        *(const bf16x8a*)(q_s + (wid * 16 + col) * 32 + kbase);
    const bf16x8a a_do =
        *(const bf16x8a*)(dO_s + (wid * 16 + col) * 32 + kbase);
    f32x4a acc_s[NF], acc_dp[NF];
#pragma unroll
    for (int f = 0; f < NF; ++f) {
      bf16x8a bk, bv;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        bk[j] = k_s[(kbase + j) * Lkp + f * 16 + col];
        bv[j] = v_s[(kbase + j) * Lkp + f * 16 + col];
      }
      acc_s[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_q, bk, (f32x4a){0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
      acc_dp[f] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          a_do, bv, (f32x4a){0.f, 0.f, 0.f, 0.f}, 0, 0, 0);
    }
#pragma unroll 4
    for (int r = 0; r < 4; ++r) {
      const int row = grp * 4 + r;
      const float m = md_s[(wid * 16 + row) * 2 + 0];
      const float invd = md_s[(wid * 16 + row) * 2 + 1];
      const float dlt = delta_s[wid * 16 + row];
#pragma unroll
      for (int f = 0; f < NF; ++f) {
        const float P = __expf(acc_s[f][r] * scale - m) * invd;
        float dP = acc_dp[f][r];
        float Pm = P;
        if (DROP) {
          const unsigned wbits = mk_s[(wid * 16 + row) * W + (f >> 1)];
          const bool keep = (wbits >> (((f & 1) << 4) + col)) & 1u;
          dP = keep ? dP * inv_keep : 0.0f;
          Pm = keep ? P * inv_keep : 0.0f;
        }
        const float dS = P * (dP - dlt);
        p_s[(wid * 16 + row) * Lkp + f * 16 + col] = (sa_bf16_a)Pm;
        ds_s[(wid * 16 + row) * Lkp + f * 16 + col] = (sa_bf16_a)dS;
      }
    }
    __syncthreads();

    // ---- dQ chunk: dq[lq, e] = scale * sum_j dS[lq,j] k[e,j] ----
    for (int ne = 0; ne < 2; ++ne) {
      if (ne >= NE) break;
      f32x4a acc_q = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int kk = 0; kk < Lk; kk += 32) {
        const bf16x8a a_ds =
            *(const bf16x8a*)(ds_s + (wid * 16 + col) * Lkp + kk + kbase);
        bf16x8a b =
            *(const bf16x8a*)(k_s + (ne * 16 + col) * Lkp + kk + kbase);
        acc_q = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_ds, b, acc_q,
                                                        0, 0, 0);
      }
      const int e = ne * 16 + col;
      if (e < E) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const long lq = lq0 + wid * 16 + grp * 4 + r;
          if (lq < Lq) {
            dq[(nh * E + e) * Lq + lq] = (sa_bf16_a)(acc_q[r] * scale);
          }
        }
      }
    }

    // ---- accumulate dV/dK tiles: A = P~^T / dS^T, B = dO / q ----
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int t = wid + 4 * i;
      if (t >= NT) break;
      const int f = t % NF;
      const int ne = t / NF;
#pragma unroll
      for (int it = 0; it < 2; ++it) {       // lq 32-chunks of the 64
        bf16x8a a_p, a_ds, b_do, b_q;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          const int lql = it * 32 + kbase + j;
          a_p[j] = p_s[lql * Lkp + f * 16 + col];
          a_ds[j] = ds_s[lql * Lkp + f * 16 + col];
          b_do[j] = dO_s[lql * 32 + ne * 16 + col];
          b_q[j] = q_s[lql * 32 + ne * 16 + col];
        }
        acc_dv[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_p, b_do, acc_dv[i], 0, 0, 0);
        acc_dk[i] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            a_ds, b_q, acc_dk[i], 0, 0, 0);
      }
    }
  }

  // ---- epilogue: dV/dK tiles (D col = e, row = key) ----
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    const int t = wid + 4 * i;
    if (t >= NT) break;
    const int f = t % NF;
    const int ne = t / NF;
    const int e = ne * 16 + col;
    if (e >= E) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int key = f * 16 + grp * 4 + r;
      dv[(nh * E + e) * Lk + key] = (sa_bf16_a)acc_dv[i][r];
      dk[(nh * E + e) * Lk + key] = (sa_bf16_a)(acc_dk[i][r] * scale);
    }
  }
}

}  // namespace

// single-launch MFMA backward; returns false outside the envelope
bool pooled_attn_bwd_mfma(const at::Tensor& q, const at::Tensor& k,
                          const at::Tensor& v, const at::Tensor& out,
                          const at::Tensor& dout, const at::Tensor& stats,
                          const at::Tensor& mask, bool drop,
                          at::Tensor& dq, at::Tensor& dk, at::Tensor& dv,
                          long NH, int E, long Lq, int Lk, float scale,
                          float inv_keep) {
  if (q.scalar_type() != at::kBFloat16) return false;
  if (Lk != 64 && Lk != 128) return false;
  if (Lq % 16 != 0) return false;
  const int Lkp = Lk + 8;
  const int W = (Lk / 16 + 1) / 2;
  const size_t lds = sizeof(sa_bf16_a)
          * (2 * 32 * Lkp + 2 * 64 * 32 + 2 * 64 * Lkp)
      + sizeof(float) * (64 * 2 + 64) + sizeof(unsigned) * 64 * W;
  dim3 grid(NH);
  auto stream = at::hip::getCurrentHIPStream();
  auto delta = at::empty({NH, Lq}, q.options().dtype(at::kFloat));
  hipLaunchKernelGGL(attn_delta_kernel,
                     dim3(sa::ceil_div(NH * Lq, (long)256)), dim3(256), 0,
                     stream.stream(), (const sa_bf16_a*)out.data_ptr(),
                     (const sa_bf16_a*)dout.data_ptr(),
                     delta.data_ptr<float>(), NH * Lq, Lq, E);
  auto launch = [&](auto nf_, auto d_) {
    hipLaunchKernelGGL(
        (pooled_attn_bwd_mfma_kernel<decltype(nf_)::value,
                                     decltype(d_)::value>),
        grid, dim3(256), lds, stream.stream(),
        (const sa_bf16_a*)q.data_ptr(), (const sa_bf16_a*)k.data_ptr(),
        (const sa_bf16_a*)v.data_ptr(), (const sa_bf16_a*)out.data_ptr(),
        (const sa_bf16_a*)dout.data_ptr(), delta.data_ptr<float>(),
        stats.data_ptr<float>(),
        drop ? (const unsigned*)mask.data_ptr<int>() : nullptr,
        (sa_bf16_a*)dq.data_ptr(), (sa_bf16_a*)dk.data_ptr(),
        (sa_bf16_a*)dv.data_ptr(), Lq, E, scale, inv_keep);
  };
  auto launch_nf = [&](auto nf_) {
    if (drop) launch(nf_, std::true_type{});
    else launch(nf_, std::false_type{});
  };
  if (Lk == 64) launch_nf(std::integral_constant<int, 4>{});
  else launch_nf(std::integral_constant<int, 8>{});
  return true;
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
  if (pooled_attn_bwd_mfma(q, k, v, out, dout, stats, mask, drop, dq, dk,
                           dv, N * H, E, Lq, Lk, scale, inv_keep)) {
    return {dq, dk, dv};
  }
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
