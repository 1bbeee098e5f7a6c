// Small-channel dense conv1d WEIGHT gradient on MFMA.
//
//   dw[co][ci][k] = sum_{n, l} dy[n][co][l] * x[n][ci][l + k*1 - padl]
//
// The per-tap bmm route (conv1d.hip) runs K batched hipblaslt GEMMs of
// shape (Co x Lo x Cig) plus a (K, N, Co*Cig) middle-axis sum; at the
// eqt/ditingmotion shapes (Co 1..16, Ci 3..64, K 3..11) that measured
// ~4-9 ms/step of GEMM + sum_mid time. Here the whole gradient is ONE
// kernel + one batch-axis sum:
//
//   D[m=co][j=(ci,tap)] = sum_l A[m][l] * B[l][j],  B[l][j] = x[ci_j][l
//   + tap_j - padl]
//
// j packs (ci, tap) DENSELY (j = ci*K + tap, no padding) as the MFMA n
// axis in JT 16-column tiles, all tiles of one (m-tile, n, l-slice) in a
// single block so dy/x stage once per 256-l chunk. The per-lane B
// fragment is 8 consecutive l of one x row at an arbitrary 2-byte offset,
// read branchlessly as 5 dwords + 4 v_alignbyte. Each wave owns a 64-l
// quarter of the chunk and writes its own fp32 slab row
// (split = ((n*lsplit + lz)*4 + wave)); sum_batch_to folds the slab.
//
// Envelope: bf16, groups == 1, stride == 1, dilation == 1, K <= 16,
// Ci <= 64, Ci*K <= 256.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

typedef __bf16 sa_bf16;
typedef sa_bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef uint32_t u32x4 __attribute__((ext_vector_type(4)));

at::Tensor sum_batch_to(const at::Tensor& in, at::ScalarType out_dtype);

namespace {

constexpr int kBlock = 256;
constexpr int kChunk = 256;     // l per block-iteration (64 per wave)
constexpr int kXExt = 280;      // staged x window (8 base + 256 + 15 tap)
constexpr int kXPitch = 288;
constexpr int kYPitch = 264;    // dy rows: 256 + 8 pad

// 8 consecutive bf16 at arbitrary 2B-aligned element offset p: 5 dword
// reads + 4 v_alignbyte with a runtime byte shift (branchless, works for
// per-lane mixed parity).
__device__ __forceinline__ bf16x8 load_row8_any(
    const uint32_t* __restrict__ xw, int p) {
  const int q = p >> 1;
  const int sh = (p & 1) * 2;
  const uint32_t w0 = xw[q], w1 = xw[q + 1], w2 = xw[q + 2],
                 w3 = xw[q + 3], w4 = xw[q + 4];
  u32x4 d;
  d[0] = __builtin_amdgcn_alignbyte(w1, w0, sh);
  d[1] = __builtin_amdgcn_alignbyte(w2, w1, sh);
  d[2] = __builtin_amdgcn_alignbyte(w3, w2, sh);
  d[3] = __builtin_amdgcn_alignbyte(w4, w3, sh);
  return __builtin_bit_cast(bf16x8, d);
}

template <int JT>
__global__ __launch_bounds__(kBlock)
void conv_dw_smallc_kernel(const sa_bf16* __restrict__ dy,
                           const sa_bf16* __restrict__ x,
                           float* __restrict__ slab,  // (N*lsplit*4, Co, CiK)
                           int N, int Ci, int Co, long Lin, long Lo,
                           int K, int padl, int lsplit, long lchunks) {
  extern __shared__ sa_bf16 smem[];
  sa_bf16* x_s = smem;                     // [Ci][kXPitch]
  sa_bf16* dy_s = smem + Ci * kXPitch;     // [16][kYPitch]

  const int mtiles = (Co + 15) / 16;
  const int m0 = ((int)blockIdx.x % mtiles) * 16;
  const int n = blockIdx.y;
  const int lz = blockIdx.z;
  const long c0 = (lchunks * lz) / lsplit;       // chunk range of this slice
  const long c1 = (lchunks * (lz + 1)) / lsplit;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int frag_m = lane & 15;
  const int kbase = (lane >> 4) * 8;
  const int CiK = Ci * K;

  const int base = (8 - (padl & 7)) & 7;  // (l0c - padl) mod 8, l0c = 256c

  // this lane's B column: j = jt*16 + (lane&15) -> (ci, tap); folded into
  // one staged-LDS element offset (row*pitch + base + tap). The row is
  // clamped so masked columns never read unstaged LDS (their D columns
  // are simply not stored).
  int pj[JT];
#pragma unroll
  for (int jt = 0; jt < JT; ++jt) {
    const int j = jt * 16 + frag_m;
    int ci = j / K;
    const int tap = j - ci * K;
    if (ci >= Ci) ci = Ci - 1;
    pj[jt] = ci * kXPitch + base + tap;
  }

  const sa_bf16* xb = x + ((long)n * Ci) * Lin;
  const sa_bf16* dyb = dy + ((long)n * Co + m0) * Lo;

  f32x4 acc[JT];
#pragma unroll
  for (int jt = 0; jt < JT; ++jt) acc[jt] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (long c = c0; c < c1; ++c) {
    const long l0 = c * kChunk;
    const long s0 = l0 - padl - base;  // 8-aligned staged x origin
    __syncthreads();
    // stage x rows [Ci][kXExt]
    for (int idx = tid; idx < Ci * (kXExt / 8); idx += kBlock) {
      const int r = idx / (kXExt / 8);
      const int e8 = idx - r * (kXExt / 8);
      const long gl = s0 + (long)e8 * 8;
      bf16x8 v = {};
      const sa_bf16* row = xb + (long)r * Lin;
      if (gl >= 0 && gl + 8 <= Lin) {
        v = *(const bf16x8*)(row + gl);
      } else {
        for (int jj = 0; jj < 8; ++jj) {
          const long lj = gl + jj;
          if (lj >= 0 && lj < Lin) v[jj] = row[lj];
        }
      }
      *(bf16x8*)(x_s + r * kXPitch + e8 * 8) = v;
    }
    // stage dy rows [16][kChunk] (zero beyond Co / Lo)
    for (int idx = tid; idx < 16 * (kChunk / 8); idx += kBlock) {
      const int r = idx / (kChunk / 8);
      const int e8 = idx - r * (kChunk / 8);
      const long gl = l0 + (long)e8 * 8;
      bf16x8 v = {};
      if (m0 + r < Co) {
        const sa_bf16* row = dyb + (long)r * Lo;
        if (gl + 8 <= Lo) {
          v = *(const bf16x8*)(row + gl);
        } else {
          for (int jj = 0; jj < 8; ++jj) {
            if (gl + jj < Lo) v[jj] = row[gl + jj];
          }
        }
      }
      *(bf16x8*)(dy_s + r * kYPitch + e8 * 8) = v;
    }
    __syncthreads();

    // each wave reduces its own 64-l quarter (2 k-steps of 32)
    const uint32_t* xw = (const uint32_t*)x_s;
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      const int lq = wid * 64 + ks * 32;
      const bf16x8 a =
          *(const bf16x8*)(dy_s + frag_m * kYPitch + lq + kbase);
#pragma unroll
      for (int jt = 0; jt < JT; ++jt) {
        const bf16x8 b = load_row8_any(xw, pj[jt] + lq + kbase);
        acc[jt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[jt],
                                                          0, 0, 0);
      }
    }
  }

  // D col = lane&15 (j), row = (lane>>4)*4 + r (m); per-wave slab row
  const int d_col = lane & 15;
  const int d_row0 = (lane >> 4) * 4;
  const long split = ((long)n * lsplit + lz) * 4 + wid;
  float* out = slab + (split * Co + m0) * CiK;
#pragma unroll
  for (int jt = 0; jt < JT; ++jt) {
    const int j = jt * 16 + d_col;
    if (j >= CiK) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int m = d_row0 + r;
      if (m0 + m < Co) out[(long)m * CiK + j] = acc[jt][r];
    }
  }
}

}  // namespace

// returns undefined tensor if the shape is outside the envelope
at::Tensor conv_dw_smallc(const at::Tensor& dy, const at::Tensor& x,
                          long K, long padl, long stride, long dilation,
                          long groups, at::ScalarType out_dtype) {
  if (dy.scalar_type() != at::kBFloat16 || x.scalar_type() != at::kBFloat16)
    return at::Tensor();
  if (groups != 1 || stride != 1 || dilation != 1) return at::Tensor();
  const int N = x.size(0), Ci = x.size(1);
  const long Lin = x.size(2);
  const int Co = dy.size(1);
  const long Lo = dy.size(2);
  const int CiK = Ci * (int)K;
  if (K < 1 || K > 16 || Ci > 64 || CiK > 256) return at::Tensor();
  static const bool off = getenv("SEIST_AMD_NO_DW_SMALLC") != nullptr;
  if (off) return at::Tensor();

  const int jt = (CiK + 15) / 16;
  const int JT = jt <= 2 ? 2 : jt <= 4 ? 4 : jt <= 6 ? 6
                 : jt <= 8 ? 8 : jt <= 12 ? 12 : 16;
  const int mtiles = (Co + 15) / 16;
  const long lchunks = (Lo + kChunk - 1) / kChunk;
  const int lsplit = (int)std::min<long>(
      lchunks, std::max<long>(1, 2048 / ((long)N * mtiles)));
  const size_t lds =
      sizeof(sa_bf16) * ((size_t)Ci * kXPitch + (size_t)16 * kYPitch);

  auto slab = at::empty({(long)N * lsplit * 4, (long)Co, (long)CiK},
                        x.options().dtype(at::kFloat));
  dim3 grid(mtiles, N, lsplit);
  auto stream = at::hip::getCurrentHIPStream();
  auto launch = [&](auto jt_c) {
    hipLaunchKernelGGL((conv_dw_smallc_kernel<decltype(jt_c)::value>),
                       grid, dim3(kBlock), lds, stream.stream(),
                       (const sa_bf16*)dy.data_ptr(),
                       (const sa_bf16*)x.data_ptr(),
                       slab.data_ptr<float>(), N, Ci, Co, Lin, Lo,
                       (int)K, (int)padl, lsplit, lchunks);
  };
  switch (JT) {
    case 2: launch(std::integral_constant<int, 2>{}); break;
    case 4: launch(std::integral_constant<int, 4>{}); break;
    case 6: launch(std::integral_constant<int, 6>{}); break;
    case 8: launch(std::integral_constant<int, 8>{}); break;
    case 12: launch(std::integral_constant<int, 12>{}); break;
    default: launch(std::integral_constant<int, 16>{}); break;
  }
  SA_CHECK_HIP(hipGetLastError());
  return sum_batch_to(slab.view({(long)N * lsplit * 4, (long)Co * CiK}),
                      out_dtype)
      .view({(long)Co, (long)Ci, (long)K});
}
