// MFMA (matrix-core) pointwise-conv GEMM for bf16 — the K1 hot op of
// SURVEY.md §2.4 on the CDNA4 matrix pipe.
//
//   y[n, m, l] = sum_k Wv[m][k] * x[n, k, l]   (+ bias[m])
//   Wv[m][k] = TRANS ? w[k*Co + m] : w[m*Ci + k]
//
// Tiling: one 256-thread block (4 waves) owns a (32 co x 128 l) output
// tile of one sample; waves split 2x2 over (co, l); each wave computes
// 16x64 via four v_mfma_f32_16x16x32_bf16 accumulators, K-stepping over
// ci in 32-wide chunks staged through LDS.
//
// Fragment maps (gfx950, 16x16x32): A[i = lane&15][k = (lane>>4)*8 + j],
// B[k = (lane>>4)*8 + j][n = lane&15], D col = lane&15,
// row = (lane>>4)*4 + reg (cdna_hip_programming.md §3).
//
// The X chunk stays row-major in LDS (coalesced b128 staging); the
// B-fragment gather pays 8 narrow LDS reads per fragment, which the four
// MFMAs per K-step hide. The W chunk is read as one contiguous bf16x8 per
// lane from a padded-row LDS image (conflict-free: 40-element row pitch).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

typedef __bf16 sa_bf16;
typedef sa_bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

at::Tensor bn_sums_only(const at::Tensor& x);

namespace {

constexpr int kBlock = 256;
constexpr int kCoT = 32;    // co per block
constexpr int kLT = 128;    // l per block
constexpr int kKT = 32;     // ci per K-step
constexpr int kWPitch = 40;   // LDS row pitch of the W chunk (bf16)
constexpr int kXPitch = 136;  // LDS row pitch of the X chunk (bf16)

// When `stats` is non-null the epilogue also accumulates per-out-channel
// (sum, sumsq) of the bf16-rounded outputs into a partial slab
// (Co, nsplit, 2), nsplit = N * gridDim.x — the producer half of the
// conv->BN chain fusion (docs/FUSION_PLAN.md step 1): the following
// BatchNorm consumes the slab and never re-reads y for its statistics.
// PRE: x rows are normalized/activated during staging
// (z = act(x * pre_scale[c] + pre_shift[c])) so the BN-apply output never
// exists in HBM — conv->BN fusion step 2 (docs/FUSION_PLAN.md).
template <bool TRANS, bool HAS_BIAS, bool PRE = false>
__global__ __launch_bounds__(kBlock)
void pw_mfma_kernel(const sa_bf16* __restrict__ x,
                    const sa_bf16* __restrict__ w,
                    const sa_bf16* __restrict__ bias,
                    sa_bf16* __restrict__ y,
                    float* __restrict__ stats,
                    const float* __restrict__ pre_scale,
                    const float* __restrict__ pre_shift, int pre_act,
                    int N, int Ci, int Co, long L) {
  __shared__ sa_bf16 w_s[kCoT * kWPitch];
  __shared__ sa_bf16 x_s[kKT * kXPitch];
  __shared__ float stats_s[kCoT * 2];

  const int n = blockIdx.y;
  const int co0 = blockIdx.z * kCoT;
  const long l0 = (long)blockIdx.x * kLT;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1;   // 0..1: co sub-block of 16
  const int wc = wid & 1;    // 0..1: l sub-chunk of 64

  const int frag_m = lane & 15;          // A row / D col index
  const int kbase = (lane >> 4) * 8;     // fragment k base

  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc2 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc3 = {0.f, 0.f, 0.f, 0.f};

  const sa_bf16* xb = x + (long)n * Ci * L;

  for (int k0 = 0; k0 < Ci; k0 += kKT) {
    // ---- stage W chunk [kCoT][kKT] (guarded, zero-padded) ----
    __syncthreads();
    for (int idx = tid; idx < kCoT * kKT; idx += kBlock) {
      const int m = idx / kKT;
      const int k = idx - m * kKT;
      const int mg = co0 + m;
      const int kg = k0 + k;
      float v = 0.0f;
      if (mg < Co && kg < Ci) {
        v = TRANS ? (float)w[(long)kg * Co + mg]
                  : (float)w[(long)mg * Ci + kg];
      }
      w_s[m * kWPitch + k] = (sa_bf16)v;
    }
    // ---- stage X chunk [kKT][kLT] row-major (b128 loads/stores) ----
    for (int idx = tid; idx < kKT * (kLT / 8); idx += kBlock) {
      const int k = idx / (kLT / 8);
      const int c8 = idx - k * (kLT / 8);
      const long lg = l0 + c8 * 8;
      const int kg = k0 + k;
      bf16x8 v = {};
      if (kg < Ci) {
        if (lg + 8 <= L) {
          v = *(const bf16x8*)(xb + (long)kg * L + lg);
        } else {
          for (int j = 0; j < 8; ++j) {
            v[j] = (lg + j < L) ? xb[(long)kg * L + lg + j] : (sa_bf16)0.f;
          }
        }
        if (PRE) {
          const float sc = pre_scale[kg];
          const float sh = pre_shift[kg];
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            v[j] = (sa_bf16)sa::act_fwd((float)v[j] * sc + sh, pre_act);
          }
        }
      }
      *(bf16x8*)(x_s + k * kXPitch + c8 * 8) = v;
    }
    __syncthreads();

    // ---- fragments + MFMA ----
    const bf16x8 a =
        *(const bf16x8*)(w_s + (wr * 16 + frag_m) * kWPitch + kbase);
    const int lb = wc * 64 + frag_m;  // this lane's l column within tile
#pragma unroll
    for (int nrep = 0; nrep < 4; ++nrep) {
      bf16x8 b;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        b[j] = x_s[(kbase + j) * kXPitch + nrep * 16 + lb];
      }
      switch (nrep) {
        case 0: acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc0, 0, 0, 0); break;
        case 1: acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc1, 0, 0, 0); break;
        case 2: acc2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc2, 0, 0, 0); break;
        case 3: acc3 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc3, 0, 0, 0); break;
      }
    }
  }

  // ---- epilogue: D col = lane&15 (l), row = (lane>>4)*4 + r (co) ----
  const int d_col = lane & 15;
  const int d_row0 = (lane >> 4) * 4;
  float ssum[4] = {0.f, 0.f, 0.f, 0.f};
  float ssum2[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int nrep = 0; nrep < 4; ++nrep) {
    const f32x4 acc = (nrep == 0) ? acc0 : (nrep == 1) ? acc1
                      : (nrep == 2) ? acc2 : acc3;
    const long lg = l0 + wc * 64 + nrep * 16 + d_col;
    if (lg >= L) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int mg = co0 + wr * 16 + d_row0 + r;
      if (mg < Co) {
        float v = acc[r];
        if (HAS_BIAS) v += (float)bias[mg];
        const sa_bf16 vb = (sa_bf16)v;
        y[((long)n * Co + mg) * L + lg] = vb;
        if (stats != nullptr) {
          const float vf = (float)vb;  // stats over the stored values
          ssum[r] += vf;
          ssum2[r] += vf * vf;
        }
      }
    }
  }
  if (stats != nullptr) {
    __syncthreads();
    for (int t = tid; t < kCoT * 2; t += kBlock) stats_s[t] = 0.0f;
    __syncthreads();
    // reduce over the 16 l-columns of each channel group, then LDS-merge
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int b = 1; b < 16; b <<= 1) {
        ssum[r] += __shfl_xor(ssum[r], b, sa::kWave);
        ssum2[r] += __shfl_xor(ssum2[r], b, sa::kWave);
      }
      if (d_col == 0) {
        const int mloc = wr * 16 + d_row0 + r;
        atomicAdd(&stats_s[mloc * 2 + 0], ssum[r]);
        atomicAdd(&stats_s[mloc * 2 + 1], ssum2[r]);
      }
    }
    __syncthreads();
    // split-major slab (nsplit, Co, 2): one coalesced 256 B run per
    // block instead of 64 scattered dwords at stride nsplit
    const long split = (long)blockIdx.y * gridDim.x + blockIdx.x;
    for (int t = tid; t < kCoT * 2; t += kBlock) {
      const int m = t >> 1;
      const int mg = co0 + m;
      if (mg < Co) {
        stats[(split * Co + mg) * 2 + (t & 1)] = stats_s[t];
      }
    }
  }
}

}  // namespace

// host entry; returns false if the shape/dtype is not handled.
// `stats_out` (optional): receives the (Co, nsplit, 2) partial-sums slab.
bool pw_mfma_gemm(const at::Tensor& x, const at::Tensor& w,
                  const c10::optional<at::Tensor>& bias, at::Tensor& y,
                  bool trans, at::Tensor* stats_out) {
  if (x.scalar_type() != at::ScalarType::BFloat16) return false;
  const int N = x.size(0), Ci = x.size(1);
  const long L = x.size(2);
  const int Co = trans ? w.size(1) : w.size(0);
  // MFMA pays off once the K dimension covers at least one 32-chunk
  // reasonably; tiny-K layers stay on the VALU kernel.
  if (Ci < 16) return false;

  const bool has_bias = bias.has_value() && bias->defined();
  at::Tensor bct;
  if (has_bias) bct = bias->to(x.scalar_type()).contiguous();

  dim3 grid(sa::ceil_div(L, kLT), N, sa::ceil_div(Co, kCoT));
  auto stream = at::hip::getCurrentHIPStream();
  const sa_bf16* xp = (const sa_bf16*)x.data_ptr();
  const sa_bf16* wp = (const sa_bf16*)w.data_ptr();
  const sa_bf16* bp = has_bias ? (const sa_bf16*)bct.data_ptr() : nullptr;
  sa_bf16* yp = (sa_bf16*)y.data_ptr();

  float* sp = nullptr;
  if (stats_out != nullptr) {
    const long nsplit = (long)N * grid.x;
    *stats_out = at::empty({nsplit, Co, 2}, x.options().dtype(at::kFloat));
    sp = stats_out->data_ptr<float>();
  }

  auto launch = [&](auto tr_, auto hb_) {
    hipLaunchKernelGGL((pw_mfma_kernel<decltype(tr_)::value,
                                       decltype(hb_)::value>),
                       grid, dim3(kBlock), 0, stream.stream(), xp, wp, bp,
                       yp, sp, nullptr, nullptr, 0, N, Ci, Co, L);
  };
  if (trans) {
    if (has_bias) launch(std::true_type{}, std::true_type{});
    else launch(std::true_type{}, std::false_type{});
  } else {
    if (has_bias) launch(std::false_type{}, std::true_type{});
    else launch(std::false_type{}, std::false_type{});
  }
  return true;
}

// ---------------------------------------------------------------------------
// Dense stride-1 Conv1d on the matrix cores: the conv is a GEMM with
// K_eff = Ci*K where the B operand rows are k-shifted x rows ("im2col in
// LDS" — nothing is materialized in HBM). Handles forward
// (y[co][l] = sum_{ci,k} w[co][ci][k] x[ci][l - padl + k*d]) and the
// input gradient (IS_DX: dx[ci][l] = sum_{co,k} w[co][ci][k]
// dy[co][l + padl - k*d]) with the same structure.
// ---------------------------------------------------------------------------

namespace {

template <bool IS_DX, bool HAS_BIAS>
__global__ __launch_bounds__(kBlock)
void conv_mfma_kernel(const sa_bf16* __restrict__ x,
                      const sa_bf16* __restrict__ w,
                      const sa_bf16* __restrict__ bias,
                      sa_bf16* __restrict__ y,
                      float* __restrict__ stats,
                      int N, int Cin, int Cout, long Lin, long Lout,
                      int K, int padl, int dil) {
  __shared__ sa_bf16 w_s[kCoT * kWPitch];
  __shared__ sa_bf16 x_s[kKT * kXPitch];
  __shared__ float stats_s[kCoT * 2];

  const int n = blockIdx.y;
  const int m0 = blockIdx.z * kCoT;
  const long l0 = (long)blockIdx.x * kLT;
  const int KK = Cin * K;  // effective GEMM K

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1;
  const int wc = wid & 1;
  const int frag_m = lane & 15;
  const int kbase = (lane >> 4) * 8;

  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc2 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc3 = {0.f, 0.f, 0.f, 0.f};

  const sa_bf16* xb = x + (long)n * Cin * Lin;

  for (int k0 = 0; k0 < KK; k0 += kKT) {
    __syncthreads();
    // ---- weight chunk [kCoT m][kKT kk] ----
    for (int idx = tid; idx < kCoT * kKT; idx += kBlock) {
      const int m = idx / kKT;
      const int kk = idx - m * kKT;
      const int mg = m0 + m;
      const int kkg = k0 + kk;
      float v = 0.0f;
      if (mg < Cout && kkg < KK) {
        if (IS_DX) {
          const int co = kkg / K;
          const int k = kkg - co * K;
          v = (float)w[((long)co * Cout + mg) * K + k];
        } else {
          v = (float)w[(long)mg * KK + kkg];
        }
      }
      w_s[m * kWPitch + kk] = (sa_bf16)v;
    }
    // ---- shifted-row chunk [kKT kk][kLT l] ----
    for (int idx = tid; idx < kKT * (kLT / 8); idx += kBlock) {
      const int kk = idx / (kLT / 8);
      const int c8 = idx - kk * (kLT / 8);
      const int kkg = k0 + kk;
      bf16x8 v = {};
      if (kkg < KK) {
        const int c = kkg / K;          // ci (fwd) or co (dx)
        const int k = kkg - c * K;
        const long shift = IS_DX ? (long)padl - (long)k * dil
                                 : (long)k * dil - (long)padl;
        const long lg = l0 + c8 * 8 + shift;
        const sa_bf16* row = xb + (long)c * Lin;
        if (lg >= 0 && lg + 8 <= Lin) {
          v = *(const bf16x8*)(row + lg);
        } else {
          for (int j = 0; j < 8; ++j) {
            const long lj = lg + j;
            v[j] = (lj >= 0 && lj < Lin) ? row[lj] : (sa_bf16)0.f;
          }
        }
      }
      *(bf16x8*)(x_s + kk * kXPitch + c8 * 8) = v;
    }
    __syncthreads();

    const bf16x8 a =
        *(const bf16x8*)(w_s + (wr * 16 + frag_m) * kWPitch + kbase);
    const int lb = wc * 64 + frag_m;
#pragma unroll
    for (int nrep = 0; nrep < 4; ++nrep) {
      bf16x8 b;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        b[j] = x_s[(kbase + j) * kXPitch + nrep * 16 + lb];
      }
      switch (nrep) {
        case 0: acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc0, 0, 0, 0); break;
        case 1: acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc1, 0, 0, 0); break;
        case 2: acc2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc2, 0, 0, 0); break;
        case 3: acc3 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc3, 0, 0, 0); break;
      }
    }
  }

  const int d_col = lane & 15;
  const int d_row0 = (lane >> 4) * 4;
  float ssum[4] = {0.f, 0.f, 0.f, 0.f};
  float ssum2[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int nrep = 0; nrep < 4; ++nrep) {
    const f32x4 acc = (nrep == 0) ? acc0 : (nrep == 1) ? acc1
                      : (nrep == 2) ? acc2 : acc3;
    const long lg = l0 + wc * 64 + nrep * 16 + d_col;
    if (lg >= Lout) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int mg = m0 + wr * 16 + d_row0 + r;
      if (mg < Cout) {
        float v = acc[r];
        if (HAS_BIAS) v += (float)bias[mg];
        const sa_bf16 vb = (sa_bf16)v;
        y[((long)n * Cout + mg) * Lout + lg] = vb;
        if (stats != nullptr) {
          const float vf = (float)vb;
          ssum[r] += vf;
          ssum2[r] += vf * vf;
        }
      }
    }
  }
  if (stats != nullptr) {
    __syncthreads();
    for (int t = tid; t < kCoT * 2; t += kBlock) stats_s[t] = 0.0f;
    __syncthreads();
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int b = 1; b < 16; b <<= 1) {
        ssum[r] += __shfl_xor(ssum[r], b, sa::kWave);
        ssum2[r] += __shfl_xor(ssum2[r], b, sa::kWave);
      }
      if (d_col == 0) {
        const int mloc = wr * 16 + d_row0 + r;
        atomicAdd(&stats_s[mloc * 2 + 0], ssum[r]);
        atomicAdd(&stats_s[mloc * 2 + 1], ssum2[r]);
      }
    }
    __syncthreads();
    const long split = (long)blockIdx.y * gridDim.x + blockIdx.x;
    for (int t = tid; t < kCoT * 2; t += kBlock) {
      const int m = t >> 1;
      const int mg = m0 + m;
      if (mg < Cout) {
        stats[(split * Cout + mg) * 2 + (t & 1)] = stats_s[t];
      }
    }
  }
}

}  // namespace


// ---------------------------------------------------------------------------
// fp32 variant on the exact f32 MFMA (v_mfma_f32_16x16x4_f32 — bitwise an
// fmaf chain, no TF32 on gfx950; ~2.4x a VALU GEMM per the CDNA4 guide).
// Fragment maps: A[i = lane&15][k = lane>>4], B[k = lane>>4][n = lane&15],
// C/D as the bf16 shapes. Same shifted-row staging as the bf16 kernel.
// ---------------------------------------------------------------------------

namespace {

constexpr int kXPitchF = 132;  // fp32 X row pitch (128 + 4)
constexpr int kWPitchF = 36;   // fp32 W row pitch (32 + 4)

template <bool IS_DX, bool HAS_BIAS>
__global__ __launch_bounds__(kBlock)
void conv_mfma_f32_kernel(const float* __restrict__ x,
                          const float* __restrict__ w,
                          const float* __restrict__ bias,
                          float* __restrict__ y,
                          int N, int Cin, int Cout, long Lin, long Lout,
                          int K, int padl, int dil) {
  __shared__ float w_s[kCoT * kWPitchF];
  __shared__ float x_s[kKT * kXPitchF];

  const int n = blockIdx.y;
  const int m0 = blockIdx.z * kCoT;
  const long l0 = (long)blockIdx.x * kLT;
  const int KK = Cin * K;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1;
  const int wc = wid & 1;
  const int frag_m = lane & 15;
  const int ksub = lane >> 4;  // 0..3 fragment k offset

  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc2 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc3 = {0.f, 0.f, 0.f, 0.f};

  const float* xb = x + (long)n * Cin * Lin;

  for (int k0 = 0; k0 < KK; k0 += kKT) {
    __syncthreads();
    for (int idx = tid; idx < kCoT * kKT; idx += kBlock) {
      const int m = idx / kKT;
      const int kk = idx - m * kKT;
      const int mg = m0 + m;
      const int kkg = k0 + kk;
      float v = 0.0f;
      if (mg < Cout && kkg < KK) {
        if (IS_DX) {
          const int co = kkg / K;
          const int k = kkg - co * K;
          v = w[((long)co * Cout + mg) * K + k];
        } else {
          v = w[(long)mg * KK + kkg];
        }
      }
      w_s[m * kWPitchF + kk] = v;
    }
    for (int idx = tid; idx < kKT * (kLT / 4); idx += kBlock) {
      const int kk = idx / (kLT / 4);
      const int c4 = idx - kk * (kLT / 4);
      const int kkg = k0 + kk;
      float4 v = {0.f, 0.f, 0.f, 0.f};
      if (kkg < KK) {
        const int c = kkg / K;
        const int k = kkg - c * K;
        const long shift = IS_DX ? (long)padl - (long)k * dil
                                 : (long)k * dil - (long)padl;
        const long lg = l0 + c4 * 4 + shift;
        const float* row = xb + (long)c * Lin;
        if (lg >= 0 && lg + 4 <= Lin) {
          v = *(const float4*)(row + lg);
        } else {
          float t[4];
          for (int j = 0; j < 4; ++j) {
            const long lj = lg + j;
            t[j] = (lj >= 0 && lj < Lin) ? row[lj] : 0.f;
          }
          v = make_float4(t[0], t[1], t[2], t[3]);
        }
      }
      *(float4*)(x_s + kk * kXPitchF + c4 * 4) = v;
    }
    __syncthreads();

    const float* wrow = w_s + (wr * 16 + frag_m) * kWPitchF;
    const int lb = wc * 64 + frag_m;
#pragma unroll
    for (int kk4 = 0; kk4 < kKT / 4; ++kk4) {
      const float a = wrow[kk4 * 4 + ksub];
      const float* xcol = x_s + (kk4 * 4 + ksub) * kXPitchF + lb;
      acc0 = __builtin_amdgcn_mfma_f32_16x16x4f32(a, xcol[0], acc0, 0, 0, 0);
      acc1 = __builtin_amdgcn_mfma_f32_16x16x4f32(a, xcol[16], acc1, 0, 0, 0);
      acc2 = __builtin_amdgcn_mfma_f32_16x16x4f32(a, xcol[32], acc2, 0, 0, 0);
      acc3 = __builtin_amdgcn_mfma_f32_16x16x4f32(a, xcol[48], acc3, 0, 0, 0);
    }
  }

  const int d_col = lane & 15;
  const int d_row0 = (lane >> 4) * 4;
#pragma unroll
  for (int nrep = 0; nrep < 4; ++nrep) {
    const f32x4 acc = (nrep == 0) ? acc0 : (nrep == 1) ? acc1
                      : (nrep == 2) ? acc2 : acc3;
    const long lg = l0 + wc * 64 + nrep * 16 + d_col;
    if (lg >= Lout) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int mg = m0 + wr * 16 + d_row0 + r;
      if (mg < Cout) {
        float v = acc[r];
        if (HAS_BIAS) v += bias[mg];
        y[((long)n * Cout + mg) * Lout + lg] = v;
      }
    }
  }
}

}  // namespace

// Dense stride-1 conv fwd/dx on MFMA; returns false if not applicable.
bool conv_mfma(const at::Tensor& x, const at::Tensor& w,
               const c10::optional<at::Tensor>& bias, at::Tensor& y,
               long padl, long dilation, bool is_dx,
               at::Tensor* stats_out) {
  const bool is_bf16 = x.scalar_type() == at::ScalarType::BFloat16;
  const bool is_f32 = x.scalar_type() == at::ScalarType::Float;
  if (!is_bf16 && !is_f32) return false;
  const int N = x.size(0), Cin = x.size(1);
  const long Lin = x.size(2);
  const int Cout = y.size(1);
  const long Lout = y.size(2);
  const int K = w.size(2);
  if (Cin * K < 32) return false;

  const bool has_bias = bias.has_value() && bias->defined();
  at::Tensor bct;
  if (has_bias) bct = bias->to(x.scalar_type()).contiguous();

  dim3 grid(sa::ceil_div(Lout, kLT), N, sa::ceil_div(Cout, kCoT));
  auto stream = at::hip::getCurrentHIPStream();

  if (is_f32) {
    const float* xp = x.data_ptr<float>();
    const float* wp = w.data_ptr<float>();
    const float* bp = has_bias ? bct.data_ptr<float>() : nullptr;
    float* yp = y.data_ptr<float>();
    auto launch = [&](auto dxp_, auto hb_) {
      hipLaunchKernelGGL((conv_mfma_f32_kernel<decltype(dxp_)::value,
                                               decltype(hb_)::value>),
                         grid, dim3(kBlock), 0, stream.stream(), xp, wp, bp,
                         yp, N, Cin, Cout, Lin, Lout, K, (int)padl,
                         (int)dilation);
    };
    if (is_dx) {
      if (has_bias) launch(std::true_type{}, std::true_type{});
      else launch(std::true_type{}, std::false_type{});
    } else {
      if (has_bias) launch(std::false_type{}, std::true_type{});
      else launch(std::false_type{}, std::false_type{});
    }
    if (stats_out != nullptr) {
      // fp32 kernel has no stats epilogue: one bn_sums pass instead
      *stats_out = bn_sums_only(y).view({1, Cout, 2});
    }
    return true;
  }

  const sa_bf16* xp = (const sa_bf16*)x.data_ptr();
  const sa_bf16* wp = (const sa_bf16*)w.data_ptr();
  const sa_bf16* bp = has_bias ? (const sa_bf16*)bct.data_ptr() : nullptr;
  sa_bf16* yp = (sa_bf16*)y.data_ptr();

  float* sp = nullptr;
  if (stats_out != nullptr) {
    const long nsplit = (long)N * grid.x;
    *stats_out = at::empty({nsplit, Cout, 2}, x.options().dtype(at::kFloat));
    sp = stats_out->data_ptr<float>();
  }

  auto launch = [&](auto dxp_, auto hb_) {
    hipLaunchKernelGGL((conv_mfma_kernel<decltype(dxp_)::value,
                                         decltype(hb_)::value>),
                       grid, dim3(kBlock), 0, stream.stream(), xp, wp, bp,
                       yp, sp, N, Cin, Cout, Lin, Lout, K, (int)padl,
                       (int)dilation);
  };
  if (is_dx) {
    if (has_bias) launch(std::true_type{}, std::true_type{});
    else launch(std::true_type{}, std::false_type{});
  } else {
    if (has_bias) launch(std::false_type{}, std::true_type{});
    else launch(std::false_type{}, std::false_type{});
  }
  return true;
}

// ---------------------------------------------------------------------------
// Multi-input / multi-output pointwise GEMM: y = W @ concat(x0,x1,x2) with
// the concat done during LDS staging (forward), and dx = W^T @ dy written
// directly into per-input contiguous tensors (backward input-gradient) —
// the stem's 3-path concat (reference models/seist.py:187-195) never
// materializes and its backward needs no narrow+copy pass.
// ---------------------------------------------------------------------------

namespace {

template <bool TRANS, bool HAS_BIAS>
__global__ __launch_bounds__(kBlock)
void pw_mfma_multi_kernel(const sa_bf16* __restrict__ x0,
                          const sa_bf16* __restrict__ x1,
                          const sa_bf16* __restrict__ x2,
                          int cb1, int cb2,     // input channel boundaries
                          const sa_bf16* __restrict__ w,
                          const sa_bf16* __restrict__ bias,
                          sa_bf16* __restrict__ y0,
                          sa_bf16* __restrict__ y1,
                          sa_bf16* __restrict__ y2,
                          int ob1, int ob2,     // output channel boundaries
                          int N, int Ci, int Co, long L) {
  __shared__ sa_bf16 w_s[kCoT * kWPitch];
  __shared__ sa_bf16 x_s[kKT * kXPitch];

  const int n = blockIdx.y;
  const int co0 = blockIdx.z * kCoT;
  const long l0 = (long)blockIdx.x * kLT;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wr = wid >> 1;
  const int wc = wid & 1;
  const int frag_m = lane & 15;
  const int kbase = (lane >> 4) * 8;

  f32x4 acc0 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc1 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc2 = {0.f, 0.f, 0.f, 0.f};
  f32x4 acc3 = {0.f, 0.f, 0.f, 0.f};

  for (int k0 = 0; k0 < Ci; k0 += kKT) {
    __syncthreads();
    for (int idx = tid; idx < kCoT * kKT; idx += kBlock) {
      const int m = idx / kKT;
      const int k = idx - m * kKT;
      const int mg = co0 + m;
      const int kg = k0 + k;
      float v = 0.0f;
      if (mg < Co && kg < Ci) {
        v = TRANS ? (float)w[(long)kg * Co + mg]
                  : (float)w[(long)mg * Ci + kg];
      }
      w_s[m * kWPitch + k] = (sa_bf16)v;
    }
    for (int idx = tid; idx < kKT * (kLT / 8); idx += kBlock) {
      const int k = idx / (kLT / 8);
      const int c8 = idx - k * (kLT / 8);
      const long lg = l0 + c8 * 8;
      const int kg = k0 + k;
      bf16x8 v = {};
      if (kg < Ci) {
        const sa_bf16* row;
        if (kg < cb1) {
          row = x0 + ((long)n * cb1 + kg) * L;
        } else if (kg < cb2) {
          row = x1 + ((long)n * (cb2 - cb1) + (kg - cb1)) * L;
        } else {
          row = x2 + ((long)n * (Ci - cb2) + (kg - cb2)) * L;
        }
        if (lg + 8 <= L) {
          v = *(const bf16x8*)(row + lg);
        } else {
          for (int j = 0; j < 8; ++j) {
            v[j] = (lg + j < L) ? row[lg + j] : (sa_bf16)0.f;
          }
        }
      }
      *(bf16x8*)(x_s + k * kXPitch + c8 * 8) = v;
    }
    __syncthreads();

    const bf16x8 a =
        *(const bf16x8*)(w_s + (wr * 16 + frag_m) * kWPitch + kbase);
    const int lb = wc * 64 + frag_m;
#pragma unroll
    for (int nrep = 0; nrep < 4; ++nrep) {
      bf16x8 b;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        b[j] = x_s[(kbase + j) * kXPitch + nrep * 16 + lb];
      }
      switch (nrep) {
        case 0: acc0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc0, 0, 0, 0); break;
        case 1: acc1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc1, 0, 0, 0); break;
        case 2: acc2 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc2, 0, 0, 0); break;
        case 3: acc3 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc3, 0, 0, 0); break;
      }
    }
  }

  const int d_col = lane & 15;
  const int d_row0 = (lane >> 4) * 4;
#pragma unroll
  for (int nrep = 0; nrep < 4; ++nrep) {
    const f32x4 acc = (nrep == 0) ? acc0 : (nrep == 1) ? acc1
                      : (nrep == 2) ? acc2 : acc3;
    const long lg = l0 + wc * 64 + nrep * 16 + d_col;
    if (lg >= L) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int mg = co0 + wr * 16 + d_row0 + r;
      if (mg < Co) {
        float v = acc[r];
        if (HAS_BIAS) v += (float)bias[mg];
        sa_bf16* dst;
        int ml, cw;
        if (mg < ob1) { dst = y0; ml = mg; cw = ob1; }
        else if (mg < ob2) { dst = y1; ml = mg - ob1; cw = ob2 - ob1; }
        else { dst = y2; ml = mg - ob2; cw = Co - ob2; }
        dst[((long)n * cw + ml) * L + lg] = (sa_bf16)v;
      }
    }
  }
}

}  // namespace

// y = W @ concat(xs) along channels; xs size 2 or 3, bf16, same (N, *, L)
at::Tensor pw_conv_multi_fwd(std::vector<at::Tensor> xs, const at::Tensor& w,
                             const c10::optional<at::Tensor>& bias) {
  TORCH_CHECK(xs.size() >= 2 && xs.size() <= 3);
  const int N = xs[0].size(0);
  const long L = xs[0].size(2);
  int Ci = 0;
  for (auto& x : xs) {
    TORCH_CHECK(x.is_cuda() && x.is_contiguous()
                && x.scalar_type() == at::kBFloat16
                && x.size(0) == N && x.size(2) == L);
    Ci += x.size(1);
  }
  const int Co = w.size(0);
  TORCH_CHECK(w.size(1) == Ci && Ci >= 16);
  auto y = at::empty({N, Co, L}, xs[0].options());
  const bool has_bias = bias.has_value() && bias->defined();
  at::Tensor bct;
  if (has_bias) bct = bias->to(at::kBFloat16).contiguous();
  const int cb1 = xs[0].size(1);
  const int cb2 = cb1 + xs[1].size(1);
  const sa_bf16* x2p = xs.size() > 2 ? (const sa_bf16*)xs[2].data_ptr()
                                     : (const sa_bf16*)xs[1].data_ptr();
  dim3 grid(sa::ceil_div(L, kLT), N, sa::ceil_div(Co, kCoT));
  auto stream = at::hip::getCurrentHIPStream();
  auto launch = [&](auto hb_) {
    hipLaunchKernelGGL((pw_mfma_multi_kernel<false, decltype(hb_)::value>),
                       grid, dim3(kBlock), 0, stream.stream(),
                       (const sa_bf16*)xs[0].data_ptr(),
                       (const sa_bf16*)xs[1].data_ptr(), x2p, cb1,
                       xs.size() > 2 ? cb2 : Ci,
                       (const sa_bf16*)w.data_ptr(),
                       has_bias ? (const sa_bf16*)bct.data_ptr() : nullptr,
                       (sa_bf16*)y.data_ptr(), (sa_bf16*)y.data_ptr(),
                       (sa_bf16*)y.data_ptr(), Co, Co, N, Ci, Co, L);
  };
  if (has_bias) launch(std::true_type{});
  else launch(std::false_type{});
  return y;
}

// dxs = split(W^T @ dy) written straight into per-input contiguous tensors
std::vector<at::Tensor> pw_conv_multi_dx(const at::Tensor& dy,
                                         const at::Tensor& w,
                                         std::vector<long> sizes) {
  TORCH_CHECK(sizes.size() >= 2 && sizes.size() <= 3);
  const int N = dy.size(0);
  const long L = dy.size(2);
  const int Cin = dy.size(1);   // GEMM reduce dim (= conv out channels)
  long Ct = 0;
  for (long s : sizes) Ct += s;
  // w is the ORIGINAL conv weight (Co_orig=Cin rows, Ci_orig=Ct cols);
  // dx[c] = sum_m w[m][c] * dy[m] -> the TRANS access w[kg * Co + mg]
  // with the kernel's Co == Ct reads w[m * Ct + c], exactly w[m][c].
  TORCH_CHECK(w.size(0) == Cin && w.size(1) == Ct);
  std::vector<at::Tensor> dxs;
  for (long s : sizes) dxs.push_back(at::empty({N, s, L}, dy.options()));
  const int ob1 = sizes[0];
  const int ob2 = ob1 + sizes[1];
  sa_bf16* y2p = sizes.size() > 2 ? (sa_bf16*)dxs[2].data_ptr()
                                  : (sa_bf16*)dxs[1].data_ptr();
  dim3 grid(sa::ceil_div(L, kLT), N, sa::ceil_div((long)Ct, (long)kCoT));
  auto stream = at::hip::getCurrentHIPStream();
  hipLaunchKernelGGL((pw_mfma_multi_kernel<true, false>), grid,
                     dim3(kBlock), 0, stream.stream(),
                     (const sa_bf16*)dy.data_ptr(),
                     (const sa_bf16*)dy.data_ptr(),
                     (const sa_bf16*)dy.data_ptr(), Cin, Cin,
                     (const sa_bf16*)w.data_ptr(), nullptr,
                     (sa_bf16*)dxs[0].data_ptr(),
                     (sa_bf16*)dxs[1].data_ptr(), y2p, ob1,
                     sizes.size() > 2 ? ob2 : (int)Ct,
                     N, Cin, (int)Ct, L);
  return dxs;
}

// ---------------------------------------------------------------------------
// BN(+act)+pointwise fusion entry points (FUSION_PLAN step 2)
// ---------------------------------------------------------------------------

// y = W @ act(x * scale[c] + shift[c]): the BN-apply output never exists
at::Tensor pw_conv_pre_fwd(const at::Tensor& x, const at::Tensor& w,
                           const c10::optional<at::Tensor>& bias,
                           const at::Tensor& scale, const at::Tensor& shift,
                           long act) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous()
              && x.scalar_type() == at::kBFloat16);
  const int N = x.size(0), Ci = x.size(1);
  const long L = x.size(2);
  const int Co = w.size(0);
  TORCH_CHECK(w.size(1) == Ci && Ci >= 16);
  TORCH_CHECK(scale.is_contiguous() && shift.is_contiguous()
              && scale.scalar_type() == at::kFloat);
  auto y = at::empty({N, Co, L}, x.options());
  const bool has_bias = bias.has_value() && bias->defined();
  at::Tensor bct;
  if (has_bias) bct = bias->to(at::kBFloat16).contiguous();
  dim3 grid(sa::ceil_div(L, kLT), N, sa::ceil_div(Co, kCoT));
  auto stream = at::hip::getCurrentHIPStream();
  auto launch = [&](auto hb_) {
    hipLaunchKernelGGL((pw_mfma_kernel<false, decltype(hb_)::value, true>),
                       grid, dim3(kBlock), 0, stream.stream(),
                       (const sa_bf16*)x.data_ptr(),
                       (const sa_bf16*)w.data_ptr(),
                       has_bias ? (const sa_bf16*)bct.data_ptr() : nullptr,
                       (sa_bf16*)y.data_ptr(), nullptr,
                       scale.data_ptr<float>(), shift.data_ptr<float>(),
                       (int)act, N, Ci, Co, L);
  };
  if (has_bias) launch(std::true_type{});
  else launch(std::false_type{});
  return y;
}

// ---------------------------------------------------------------------------
// split-K weight gradient with the BN transform applied while staging x:
//   dw[m][c] = sum_{n,l} dy[n,m,l] * act(x[n,c,l]*scale[c]+shift[c])
// One block per (32m x 32c tile, n); 2x2 waves, each a 16x16 MFMA tile,
// K-stepping l in 32-chunks. Partials land in a per-sample fp32 slab
// summed by sum_batch (same reduction the hipblaslt bmm path uses).
// ---------------------------------------------------------------------------

namespace {

constexpr int kDwL = 32;  // l per K-step

template <bool PRE>
__global__ __launch_bounds__(kBlock)
void pw_dw_kernel(const sa_bf16* __restrict__ dy,
                  const sa_bf16* __restrict__ x,
                  const float* __restrict__ pre_scale,
                  const float* __restrict__ pre_shift, int pre_act,
                  float* __restrict__ slab,  // (N, Co, Ci)
                  int N, int Co, int Ci, long L) {
  __shared__ sa_bf16 dy_s[32 * 40];   // [32 m][32 l] padded
  __shared__ sa_bf16 x_s[32 * 40];    // [32 c][32 l] padded
  const int n = blockIdx.y;
  const int mtiles = (Co + 31) / 32;
  const int m0 = (blockIdx.x % mtiles) * 32;
  const int c0 = (blockIdx.x / mtiles) * 32;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int wm = wid >> 1;   // 16-m sub-tile
  const int wc = wid & 1;    // 16-c sub-tile
  const int frag_m = lane & 15;
  const int kbase = (lane >> 4) * 8;

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  const sa_bf16* dyb = dy + ((long)n * Co) * L;
  const sa_bf16* xb = x + ((long)n * Ci) * L;

  for (long l0 = 0; l0 < L; l0 += kDwL) {
    __syncthreads();
    // stage dy rows [32 m][32 l] and x rows [32 c][32 l]
    for (int idx = tid; idx < 32 * (kDwL / 8); idx += kBlock) {
      const int r = idx / (kDwL / 8);
      const int c8 = idx - r * (kDwL / 8);
      const long lg = l0 + c8 * 8;
      bf16x8 vd = {}, vx = {};
      if (m0 + r < Co) {
        const sa_bf16* row = dyb + (long)(m0 + r) * L;
        if (lg + 8 <= L) vd = *(const bf16x8*)(row + lg);
        else for (int j = 0; j < 8; ++j)
          vd[j] = (lg + j < L) ? row[lg + j] : (sa_bf16)0.f;
      }
      if (c0 + r < Ci) {
        const sa_bf16* row = xb + (long)(c0 + r) * L;
        if (lg + 8 <= L) vx = *(const bf16x8*)(row + lg);
        else for (int j = 0; j < 8; ++j)
          vx[j] = (lg + j < L) ? row[lg + j] : (sa_bf16)0.f;
        if (PRE) {
          const float sc = pre_scale[c0 + r];
          const float sh = pre_shift[c0 + r];
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            vx[j] = (sa_bf16)sa::act_fwd((float)vx[j] * sc + sh, pre_act);
          }
        }
      }
      *(bf16x8*)(dy_s + r * 40 + c8 * 8) = vd;
      *(bf16x8*)(x_s + r * 40 + c8 * 8) = vx;
    }
    __syncthreads();

    // A[i=m][k=l] from dy_s; B[k=l][j=c] = z[c=j][l=k] — both contiguous
    // 8-element runs because the LDS images are [row][l]
    const bf16x8 a = *(const bf16x8*)(dy_s + (wm * 16 + frag_m) * 40 + kbase);
    const bf16x8 b =
        *(const bf16x8*)(x_s + (wc * 16 + frag_m) * 40 + kbase);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }

  // D col = lane&15 (c), row = (lane>>4)*4 + r (m)
  const int d_col = lane & 15;
  const int d_row0 = (lane >> 4) * 4;
  const int cg = c0 + wc * 16 + d_col;
  if (cg >= Ci) return;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int mg = m0 + wm * 16 + d_row0 + r;
    if (mg < Co) {
      slab[((long)n * Co + mg) * Ci + cg] = acc[r];
    }
  }
}

}  // namespace

at::Tensor sum_batch(const at::Tensor& in);
at::Tensor sum_batch_to(const at::Tensor& in, at::ScalarType out_dtype);

// dw = sum_n dy_n @ act(x_n * scale + shift)^T  (fp32 out)
at::Tensor pw_dw_pre(const at::Tensor& dy, const at::Tensor& x,
                     const c10::optional<at::Tensor>& scale,
                     const c10::optional<at::Tensor>& shift, long act,
                     c10::optional<at::ScalarType> out_dtype) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  const int N = dy.size(0), Co = dy.size(1);
  const long L = dy.size(2);
  const int Ci = x.size(1);
  auto slab = at::empty({N, Co, Ci}, dy.options().dtype(at::kFloat));
  const int mtiles = (Co + 31) / 32;
  const int ctiles = (Ci + 31) / 32;
  dim3 grid(mtiles * ctiles, N);
  auto stream = at::hip::getCurrentHIPStream();
  const bool pre = scale.has_value() && scale->defined();
  if (pre) {
    hipLaunchKernelGGL((pw_dw_kernel<true>), grid, dim3(kBlock), 0,
                       stream.stream(), (const sa_bf16*)dy.data_ptr(),
                       (const sa_bf16*)x.data_ptr(),
                       scale->data_ptr<float>(), shift->data_ptr<float>(),
                       (int)act, slab.data_ptr<float>(), N, Co, Ci, L);
  } else {
    hipLaunchKernelGGL((pw_dw_kernel<false>), grid, dim3(kBlock), 0,
                       stream.stream(), (const sa_bf16*)dy.data_ptr(),
                       (const sa_bf16*)x.data_ptr(), nullptr, nullptr, 0,
                       slab.data_ptr<float>(), N, Co, Ci, L);
  }
  return sum_batch_to(slab, out_dtype.value_or(at::kFloat));
}
