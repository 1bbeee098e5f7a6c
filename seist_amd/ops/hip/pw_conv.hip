// Pointwise (1x1) Conv1d as a channels-first batched GEMM — K1 of
// SURVEY.md §2.4 (replaces the reference's ubiquitous 1x1 nn.Conv1d,
// models/seist.py:106-113 and friends).
//
// Layout: x (N, Ci, L) contiguous, w (Co, Ci), y (N, Co, L).
// y[n][o][l] = sum_i w[o][i] * x[n][i][l] (+ b[o])
//
// Design (memory-first, CDNA4): the weight panel is tiny (Ci,Co <= ~400)
// and lives in LDS; each 256-thread block owns 256 consecutive `l`
// positions of one (n, co-chunk) tile, so every global read/write is a
// fully coalesced 64-lane row and x is re-read only ceil(Co/32) times
// (L2-resident between passes). Accumulation fp32.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kCoChunk = 32;

// flag TRANS: W stored (I, O) instead of (O, I) — used by dx = W^T @ dy.
template <typename scalar_t, bool TRANS, bool HAS_BIAS>
__global__ void pw_gemm_kernel(const scalar_t* __restrict__ x,
                               const scalar_t* __restrict__ w,
                               const float* __restrict__ bias,
                               scalar_t* __restrict__ y,
                               int N, int Ci, int Co, long L) {
  __shared__ float w_lds[kCoChunk * 129];  // [co][ci] chunk, padded stride

  const int n = blockIdx.y;
  const int co0 = blockIdx.z * kCoChunk;
  const int co_n = min(kCoChunk, Co - co0);
  const long l = (long)blockIdx.x * kBlock + threadIdx.x;

  float acc[kCoChunk];
#pragma unroll
  for (int j = 0; j < kCoChunk; ++j) acc[j] = 0.0f;

  for (int ci0 = 0; ci0 < Ci; ci0 += 128) {
    const int ci_n = min(128, Ci - ci0);
    __syncthreads();
    for (int idx = threadIdx.x; idx < co_n * ci_n; idx += kBlock) {
      const int jo = idx / ci_n;
      const int ji = idx - jo * ci_n;
      const int o = co0 + jo;
      const int i = ci0 + ji;
      w_lds[jo * 129 + ji] =
          TRANS ? (float)w[(long)i * Co + o] : (float)w[(long)o * Ci + i];
    }
    __syncthreads();
    if (l < L) {
      const scalar_t* xp = x + ((long)n * Ci + ci0) * L + l;
      for (int ji = 0; ji < ci_n; ++ji) {
        const float xv = (float)xp[(long)ji * L];
#pragma unroll
        for (int jo = 0; jo < kCoChunk; ++jo) {
          acc[jo] += w_lds[jo * 129 + ji] * xv;
        }
      }
    }
  }

  if (l < L) {
    scalar_t* yp = y + ((long)n * Co + co0) * L + l;
    for (int jo = 0; jo < co_n; ++jo) {
      float v = acc[jo];
      if (HAS_BIAS) v += bias[co0 + jo];
      yp[(long)jo * L] = (scalar_t)v;
    }
  }
}

// dw[o][i] = sum_{n,l} dy[n][o][l] * x[n][i][l] — a (Co x Ci) GEMM whose
// K dimension is the flattened (n,l) axis. 16x16 output tile per block,
// one (o,i) pair per thread, K staged through LDS in 64-wide slabs,
// split-K over blockIdx.z with fp32 atomics. db folded in.
template <typename scalar_t, bool HAS_BIAS>
__global__ void pw_dw_kernel(const scalar_t* __restrict__ dy,
                             const scalar_t* __restrict__ x,
                             float* __restrict__ dw,
                             float* __restrict__ db,
                             int N, int Ci, int Co, long L, int nsplit) {
  constexpr int T = 16;   // tile side; T*T == blockDim.x
  constexpr int KS = 64;  // k-slab width
  __shared__ float dy_s[T][KS + 1];
  __shared__ float x_s[T][KS + 1];

  const int o0 = blockIdx.x * T;
  const int i0 = blockIdx.y * T;
  const long total = (long)N * L;
  const long chunk = (total + nsplit - 1) / nsplit;
  const long k0 = (long)blockIdx.z * chunk;
  const long k1 = min(total, k0 + chunk);

  const int po = threadIdx.x / T;
  const int pi = threadIdx.x % T;
  const bool valid = (o0 + po < Co) && (i0 + pi < Ci);

  float acc = 0.0f;
  float bacc = 0.0f;

  for (long ks = k0; ks < k1; ks += KS) {
    const int kn = (int)min((long)KS, k1 - ks);
    // stage dy rows and x rows: 4 rows per wave pass
    __syncthreads();
    for (int idx = threadIdx.x; idx < T * KS; idx += kBlock) {
      const int r = idx / KS;
      const int kk = idx % KS;
      const long k = ks + kk;
      float dv = 0.0f, xv = 0.0f;
      if (kk < kn && k < total) {
        const long n = k / L;
        const long l = k - n * L;
        if (o0 + r < Co) dv = (float)dy[((long)n * Co + o0 + r) * L + l];
        if (i0 + r < Ci) xv = (float)x[((long)n * Ci + i0 + r) * L + l];
      }
      dy_s[r][kk] = dv;
      x_s[r][kk] = xv;
    }
    __syncthreads();
#pragma unroll 16
    for (int kk = 0; kk < KS; ++kk) {
      acc += dy_s[po][kk] * x_s[pi][kk];
      // db only once per (o, k) — not per ci-tile
      if (HAS_BIAS && pi == 0 && blockIdx.y == 0) bacc += dy_s[po][kk];
    }
  }

  if (valid) atomicAdd(&dw[(long)(o0 + po) * Ci + i0 + pi], acc);
  if (HAS_BIAS && pi == 0 && blockIdx.y == 0 && o0 + po < Co) {
    atomicAdd(&db[o0 + po], bacc);
  }
}

}  // namespace

at::Tensor pw_conv_fwd(const at::Tensor& x, const at::Tensor& w,
                       const c10::optional<at::Tensor>& bias) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  TORCH_CHECK(w.is_cuda() && w.is_contiguous() && w.dim() == 2);
  const int N = x.size(0), Ci = x.size(1);
  const long L = x.size(2);
  const int Co = w.size(0);
  TORCH_CHECK(w.size(1) == Ci, "weight/input channel mismatch");
  auto y = at::empty({N, Co, L}, x.options());

  at::Tensor b32;
  const bool has_bias = bias.has_value() && bias->defined();
  if (has_bias) b32 = bias->to(at::kFloat).contiguous();

  dim3 grid(sa::ceil_div(L, kBlock), N, sa::ceil_div(Co, kCoChunk));
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "pw_conv_fwd", [&] {
        if (has_bias) {
          hipLaunchKernelGGL((pw_gemm_kernel<scalar_t, false, true>), grid,
                             dim3(kBlock), 0, stream.stream(),
                             x.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                             b32.data_ptr<float>(), y.data_ptr<scalar_t>(),
                             N, Ci, Co, L);
        } else {
          hipLaunchKernelGGL((pw_gemm_kernel<scalar_t, false, false>), grid,
                             dim3(kBlock), 0, stream.stream(),
                             x.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                             nullptr, y.data_ptr<scalar_t>(),
                             N, Ci, Co, L);
        }
      });
  return y;
}

std::vector<at::Tensor> pw_conv_bwd(const at::Tensor& dy, const at::Tensor& x,
                                    const at::Tensor& w, bool has_bias) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous());
  const int N = x.size(0), Ci = x.size(1);
  const long L = x.size(2);
  const int Co = w.size(0);
  auto stream = at::hip::getCurrentHIPStream();

  // dx = W^T @ dy (same kernel, transposed weight view)
  auto dx = at::empty_like(x);
  {
    dim3 grid(sa::ceil_div(L, kBlock), N, sa::ceil_div(Ci, kCoChunk));
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
        "pw_conv_dx", [&] {
          hipLaunchKernelGGL((pw_gemm_kernel<scalar_t, true, false>), grid,
                             dim3(kBlock), 0, stream.stream(),
                             dy.data_ptr<scalar_t>(), w.data_ptr<scalar_t>(),
                             nullptr, dx.data_ptr<scalar_t>(),
                             N, Co, Ci, L);
        });
  }

  auto dw32 = at::zeros({Co, Ci}, x.options().dtype(at::kFloat));
  at::Tensor db32;
  if (has_bias) db32 = at::zeros({Co}, x.options().dtype(at::kFloat));
  {
    const int nsplit = std::max(
        1, std::min<int>(128, (int)(((long)N * L) / 65536) + 1));
    dim3 grid(sa::ceil_div(Co, 16), sa::ceil_div(Ci, 16), nsplit);
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
        "pw_conv_dw", [&] {
          if (has_bias) {
            hipLaunchKernelGGL((pw_dw_kernel<scalar_t, true>), grid,
                               dim3(kBlock), 0, stream.stream(),
                               dy.data_ptr<scalar_t>(), x.data_ptr<scalar_t>(),
                               dw32.data_ptr<float>(), db32.data_ptr<float>(),
                               N, Ci, Co, L, nsplit);
          } else {
            hipLaunchKernelGGL((pw_dw_kernel<scalar_t, false>), grid,
                               dim3(kBlock), 0, stream.stream(),
                               dy.data_ptr<scalar_t>(), x.data_ptr<scalar_t>(),
                               dw32.data_ptr<float>(), nullptr,
                               N, Ci, Co, L, nsplit);
          }
        });
  }
  auto dw = dw32.to(w.scalar_type());
  at::Tensor db;
  if (has_bias) db = db32.to(w.scalar_type());
  return {dx, dw, db};
}
