// Small-channel dense conv1d forward / input-gradient on MFMA (K4/K6).
//
// The tap-gather kernel (conv_tap.hip) tiles the reduce dimension in
// 32-channel chunks and the m axis in 16 output channels. At the
// eqt/phasenet/ditingmotion shapes (Ci 3..32, Co 1..16, K 3..11, L up to
// 8192) that wastes most of the matrix pipe (Ci=16 fills half a chunk,
// Co=8 half a tile) and its B fragments are gathered with 8 scalar u16
// LDS reads per MFMA — measured 6-15x off the HBM roofline.
//
// This kernel packs the reduce dimension DENSELY as k = ci*KP + tap
// (KP = 8 or 16, taps zero-padded in the weight image), so
//   * the reduce width is Ci*KP instead of ceil(Ci/32)*32*K,
//   * the 8 consecutive k values a B fragment lane needs are 8 CONSECUTIVE
//     taps of one channel row = 8 consecutive staged x elements
//     (dilation 1), loaded as 4 dword LDS reads (even offset) or 5 dwords
//     + 4 v_alignbit (odd offset) instead of 8 scalar reads.
// Output columns map l in PAIRS (two 16-column sub-lattices, even and odd
// l), so each lane stores one packed dword of two bf16 — fully coalesced.
// The whole x window (<= 32 rows x 288) and the zero-padded weight image
// stage once per block; there is no chunk loop.
//
// Envelope: bf16, groups == 1, stride == 1, dilation == 1, K <= 16,
// reduce channels <= 32, Lout even. Fragment maps as in pw_mfma.hip
// (gfx950 v_mfma_f32_16x16x32_bf16).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

typedef __bf16 sa_bf16;
typedef sa_bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));
typedef uint32_t u32x4 __attribute__((ext_vector_type(4)));

namespace {

constexpr int kBlock = 256;
constexpr int kLT = 256;          // l per block (4 waves x 64)
constexpr int kXExt = 280;        // staged window: 8(base) + 15(tap) + 256 + 1
constexpr int kXPitch = 288;      // row pitch (dword-even, bank-spread)
constexpr int kMaxLds = 56 * 1024;

// 8 consecutive bf16 from a staged row at arbitrary element offset p >= 0.
// Even p: 4 aligned dword reads. Odd p: 5 dwords re-packed with alignbit.
__device__ __forceinline__ bf16x8 load_row8(const uint32_t* __restrict__ xw,
                                            int p) {
  u32x4 d;
  if ((p & 1) == 0) {
    const int q = p >> 1;
    d[0] = xw[q];
    d[1] = xw[q + 1];
    d[2] = xw[q + 2];
    d[3] = xw[q + 3];
  } else {
    const int q = (p - 1) >> 1;
    const uint32_t w0 = xw[q], w1 = xw[q + 1], w2 = xw[q + 2],
                   w3 = xw[q + 3], w4 = xw[q + 4];
    d[0] = __builtin_amdgcn_alignbit(w1, w0, 16);
    d[1] = __builtin_amdgcn_alignbit(w2, w1, 16);
    d[2] = __builtin_amdgcn_alignbit(w3, w2, 16);
    d[3] = __builtin_amdgcn_alignbit(w4, w3, 16);
  }
  return __builtin_bit_cast(bf16x8, d);
}

template <bool IS_DX, bool HAS_BIAS, int KP>
__global__ __launch_bounds__(kBlock)
void conv_smallc_kernel(const sa_bf16* __restrict__ x,
                        const sa_bf16* __restrict__ w,
                        const sa_bf16* __restrict__ bias,
                        sa_bf16* __restrict__ y,
                        float* __restrict__ stats,
                        int N, int CiR, int Cm, long Lin, long Lout,
                        int K, int padl, int NK, int CiRpad, int wpitch) {
  extern __shared__ sa_bf16 smem[];
  sa_bf16* w_s = smem;                 // [16][wpitch]  dense (ci*KP+tap) image
  sa_bf16* x_s = smem + 16 * wpitch;   // [CiRpad][kXPitch]
  __shared__ float stats_s[16 * 2];

  const int n = blockIdx.y;
  const int m0 = blockIdx.z * 16;
  const long l0 = (long)blockIdx.x * kLT;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int frag_m = lane & 15;       // A row / D column index
  const int kbase = (lane >> 4) * 8;  // fragment k offset within a kstep

  // ---- stage the zero-padded dense weight image [16 m][NK*32 k] ----
  const int KR = NK * 32;
  for (int idx = tid; idx < 16 * KR; idx += kBlock) {
    const int m = idx / KR;
    const int k = idx - m * KR;
    const int ci = k / KP;
    const int tap = k - ci * KP;
    float v = 0.0f;
    const int mg = m0 + m;
    if (mg < Cm && ci < CiR && tap < K) {
      // conv-space roles: for dx the x param is dy (reduce over conv
      // output channels) and taps run reversed
      if (IS_DX) {
        v = (float)w[((long)ci * Cm + mg) * K + (K - 1 - tap)];
      } else {
        v = (float)w[((long)mg * CiR + ci) * K + tap];
      }
    }
    w_s[m * wpitch + k] = (sa_bf16)v;
  }

  // ---- stage the x window [CiRpad rows][kXExt], aligned b128 loads ----
  const long min_off = IS_DX ? (long)padl - (long)(K - 1) : -(long)padl;
  const long s0 = (l0 + min_off) & ~7L;
  const int base = (int)(l0 + min_off - s0);  // in [0, 8)
  const sa_bf16* xb = x + (long)n * CiR * Lin;
  for (int idx = tid; idx < CiRpad * (kXExt / 8); idx += kBlock) {
    const int c = idx / (kXExt / 8);
    const int e8 = idx - c * (kXExt / 8);
    const long gl = s0 + (long)e8 * 8;
    bf16x8 v = {};
    if (c < CiR) {
      const sa_bf16* row = xb + (long)c * Lin;
      if (gl >= 0 && gl + 8 <= Lin) {
        v = *(const bf16x8*)(row + gl);
      } else {
        for (int j = 0; j < 8; ++j) {
          const long lj = gl + j;
          if (lj >= 0 && lj < Lin) v[j] = row[lj];
        }
      }
    }
    *(bf16x8*)(x_s + c * kXPitch + e8 * 8) = v;
  }
  __syncthreads();

  // lane's B row and tap base inside a kstep (dense packing)
  const int row_in = (KP == 16) ? (lane >> 5) : (lane >> 4);
  const int row_step = 32 / KP;  // reduce-channel rows consumed per kstep
  const int ktp0 = (KP == 16) ? ((lane >> 4) & 1) * 8 : 0;

  f32x4 accE0 = {0.f, 0.f, 0.f, 0.f};  // group 0, even-l sub-lattice
  f32x4 accO0 = {0.f, 0.f, 0.f, 0.f};
  f32x4 accE1 = {0.f, 0.f, 0.f, 0.f};  // group 1
  f32x4 accO1 = {0.f, 0.f, 0.f, 0.f};

  const int p00 = base + ktp0 + wid * 64 + 2 * frag_m;       // group 0 col
  for (int c = 0; c < NK; ++c) {
    const bf16x8 a =
        *(const bf16x8*)(w_s + frag_m * wpitch + c * 32 + kbase);
    const uint32_t* xw = (const uint32_t*)
        (x_s + (c * row_step + row_in) * kXPitch);
    bf16x8 b;
    b = load_row8(xw, p00);
    accE0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, accE0, 0, 0, 0);
    b = load_row8(xw, p00 + 1);
    accO0 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, accO0, 0, 0, 0);
    b = load_row8(xw, p00 + 32);
    accE1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, accE1, 0, 0, 0);
    b = load_row8(xw, p00 + 33);
    accO1 = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, accO1, 0, 0, 0);
  }

  // ---- epilogue: lane packs (even l, odd l) into one dword store ----
  // D fragment: col = lane&15 (l pair index), row = (lane>>4)*4 + r (m)
  const int d_col = lane & 15;
  const int d_row0 = (lane >> 4) * 4;
  float ssum[4] = {0.f, 0.f, 0.f, 0.f};
  float ssum2[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
  for (int grp = 0; grp < 2; ++grp) {
    const f32x4 accE = grp == 0 ? accE0 : accE1;
    const f32x4 accO = grp == 0 ? accO0 : accO1;
    const long lg = l0 + wid * 64 + grp * 32 + 2 * d_col;  // even
    if (lg >= Lout) continue;
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int mg = m0 + d_row0 + r;
      if (mg < Cm) {
        float ve = accE[r], vo = accO[r];
        if (HAS_BIAS) {
          const float bv = (float)bias[mg];
          ve += bv;
          vo += bv;
        }
        const sa_bf16 vbe = (sa_bf16)ve;
        const sa_bf16 vbo = (sa_bf16)vo;
        // Lout is even and lg even, so lg+1 < Lout and the dword store
        // at element lg is 4-byte aligned
        sa_bf16* out = y + ((long)n * Cm + mg) * Lout + lg;
        uint32_t pack;
        {
          const uint16_t lo = __builtin_bit_cast(uint16_t, vbe);
          const uint16_t hi = __builtin_bit_cast(uint16_t, vbo);
          pack = (uint32_t)lo | ((uint32_t)hi << 16);
        }
        *(uint32_t*)out = pack;
        if (stats != nullptr) {
          const float fe = (float)vbe, fo = (float)vbo;
          ssum[r] += fe + fo;
          ssum2[r] += fe * fe + fo * fo;
        }
      }
    }
  }

  if (stats != nullptr) {
    __syncthreads();
    for (int t = tid; t < 16 * 2; t += kBlock) stats_s[t] = 0.0f;
    __syncthreads();
#pragma unroll
    for (int r = 0; r < 4; ++r) {
#pragma unroll
      for (int b = 1; b < 16; b <<= 1) {
        ssum[r] += __shfl_xor(ssum[r], b, sa::kWave);
        ssum2[r] += __shfl_xor(ssum2[r], b, sa::kWave);
      }
      if (d_col == 0) {
        atomicAdd(&stats_s[(d_row0 + r) * 2 + 0], ssum[r]);
        atomicAdd(&stats_s[(d_row0 + r) * 2 + 1], ssum2[r]);
      }
    }
    __syncthreads();
    const long split = (long)blockIdx.y * gridDim.x + blockIdx.x;
    for (int t = tid; t < 16 * 2; t += kBlock) {
      const int mg = m0 + (t >> 1);
      if (mg < Cm) {
        stats[(split * Cm + mg) * 2 + (t & 1)] = stats_s[t];
      }
    }
  }
}

}  // namespace

// returns false if the shape/dtype is outside this kernel's envelope
bool conv_smallc_mfma(const at::Tensor& x, const at::Tensor& w,
                      const c10::optional<at::Tensor>& bias, at::Tensor& y,
                      long padl, long dilation, long groups, bool is_dx,
                      at::Tensor* stats_out) {
  if (x.scalar_type() != at::kBFloat16 || w.scalar_type() != at::kBFloat16)
    return false;
  if (groups != 1 || dilation != 1) return false;
  const int N = x.size(0), CiR = x.size(1);
  const long Lin = x.size(2);
  const int Cm = y.size(1);
  const long Lout = y.size(2);
  const int K = w.size(2);
  // K=1 is a pointwise GEMM: the KP=8 tap padding would waste 7/8 of the
  // reduce — leave it to the tap/im2col paths
  if (K < 2 || K > 16) return false;
  if (CiR > 32) return false;
  if (Lout & 1) return false;  // packed dword stores need even rows
  static const bool off = getenv("SEIST_AMD_NO_SMALLC") != nullptr;
  if (off) return false;

  const int KP = (K <= 8) ? 8 : 16;
  const int NK = (CiR * KP + 31) / 32;
  const int CiRpad = (NK * 32 + KP - 1) / KP;
  const int wpitch = NK * 32 + 8;
  const size_t lds =
      sizeof(sa_bf16) * ((size_t)16 * wpitch + (size_t)CiRpad * kXPitch);
  if (lds > kMaxLds) return false;

  dim3 grid(sa::ceil_div(Lout, (long)kLT), N, (Cm + 15) / 16);
  auto stream = at::hip::getCurrentHIPStream();
  const bool has_bias = bias.has_value() && bias->defined();
  at::Tensor bct;
  if (has_bias) bct = bias->to(x.scalar_type()).contiguous();
  const sa_bf16* bp = has_bias ? (const sa_bf16*)bct.data_ptr() : nullptr;

  float* sp = nullptr;
  if (stats_out != nullptr) {
    const long nsplit = (long)N * grid.x;
    *stats_out = at::empty({nsplit, Cm, 2}, x.options().dtype(at::kFloat));
    sp = stats_out->data_ptr<float>();
  }

  auto launch = [&](auto dx_t, auto hb_t, auto kp_t) {
    hipLaunchKernelGGL(
        (conv_smallc_kernel<decltype(dx_t)::value, decltype(hb_t)::value,
                            decltype(kp_t)::value>),
        grid, dim3(kBlock), lds, stream.stream(),
        (const sa_bf16*)x.data_ptr(), (const sa_bf16*)w.data_ptr(), bp,
        (sa_bf16*)y.data_ptr(), sp, N, CiR, Cm, Lin, Lout, K, (int)padl,
        NK, CiRpad, wpitch);
  };
  auto with_kp = [&](auto dx_t, auto hb_t) {
    if (KP == 8) launch(dx_t, hb_t, std::integral_constant<int, 8>{});
    else launch(dx_t, hb_t, std::integral_constant<int, 16>{});
  };
  if (is_dx) {
    if (has_bias) with_kp(std::true_type{}, std::true_type{});
    else with_kp(std::true_type{}, std::false_type{});
  } else {
    if (has_bias) with_kp(std::false_type{}, std::true_type{});
    else with_kp(std::false_type{}, std::false_type{});
  }
  SA_CHECK_HIP(hipGetLastError());
  return true;
}
