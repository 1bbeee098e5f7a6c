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

at::Tensor sum_batch(const at::Tensor& in);
at::Tensor sum_batch_to(const at::Tensor& in, at::ScalarType out_dtype);
at::Tensor channel_sum(const at::Tensor& in);
at::Tensor channel_sum_to(const at::Tensor& in, at::ScalarType out_dtype);
at::Tensor bn_sums_only(const at::Tensor& x);
bool pw_mfma_gemm(const at::Tensor& x, const at::Tensor& w,
                  const c10::optional<at::Tensor>& bias, at::Tensor& y,
                  bool trans, at::Tensor* stats_out = nullptr);
at::Tensor pw_dw_pre(const at::Tensor& dy, const at::Tensor& x,
                     const c10::optional<at::Tensor>& scale,
                     const c10::optional<at::Tensor>& shift, long act,
                     c10::optional<at::ScalarType> out_dtype);

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

  // bf16 path runs on the matrix cores
  if (pw_mfma_gemm(x, w, bias, y, /*trans=*/false)) return y;

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

// forward + per-out-channel (sum, sumsq) partials for the following
// BatchNorm (conv->BN fusion step 1, docs/FUSION_PLAN.md): the stats are
// reduced in the conv epilogue while the tile is still in registers, so
// bn_sums never re-reads y.
std::vector<at::Tensor> pw_conv_fwd_stats(
    const at::Tensor& x, const at::Tensor& w,
    const c10::optional<at::Tensor>& bias) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int N = x.size(0);
  const long L = x.size(2);
  const int Co = w.size(0);
  auto y = at::empty({N, Co, L}, x.options());
  at::Tensor part;
  if (pw_mfma_gemm(x, w, bias, y, /*trans=*/false, &part)) {
    return {y, part};
  }
  y = pw_conv_fwd(x, w, bias);
  part = bn_sums_only(y).view({1, Co, 2});
  return {y, part};
}

// input-gradient only (dx = W^T @ dy) — used by the fused BN+act+conv op
at::Tensor pw_conv_dx(const at::Tensor& dy, const at::Tensor& w) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous());
  const int N = dy.size(0);
  const long L = dy.size(2);
  const int Ci = w.size(1);
  auto dx = at::empty({N, Ci, L}, dy.options());
  TORCH_CHECK(pw_mfma_gemm(dy, w, c10::nullopt, dx, /*trans=*/true),
              "pw_conv_dx: bf16 MFMA path required");
  return dx;
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
  if (!pw_mfma_gemm(dy, w, c10::nullopt, dx, /*trans=*/true)) {
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

  // dw[o][i] = sum_n dy_n (Co x L) @ x_n^T (L x Ci). Square-ish shapes
  // go to the in-tree split-K MFMA kernel (measured 20-25 us vs
  // hipblaslt+sum_batch's 28 us at 32..96 channels); rectangular MLP
  // shapes stay on the library GEMM, which wins there.
  at::Tensor dw;
  const int Cmax = std::max(Co, Ci), Cmin = std::min(Co, Ci);
  if (dy.scalar_type() == at::kBFloat16 && Cmax <= 128 && Cmin >= 32) {
    dw = pw_dw_pre(dy, x, c10::nullopt, c10::nullopt, 0,
                   w.scalar_type());
  } else {
    dw = sum_batch_to(at::bmm(dy, x.transpose(1, 2)), w.scalar_type());
  }
  at::Tensor db;
  if (has_bias) {
    db = channel_sum_to(dy, w.scalar_type());
  }
  return {dx, dw, db};
}
