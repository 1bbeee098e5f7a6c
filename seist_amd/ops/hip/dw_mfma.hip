// MFMA weight-gradient for stride-1/2 grouped/depthwise 1-D convs.
//
//   dw[co, cig, k] = sum_{n, lo} dy[n, co, lo] * x[n, ci(co,cig), lo + k - padl]
//
// The contraction over (n, lo) is a GEMM with M = co, N = ci, K = n*lo, so
// it belongs on the matrix cores even when only a co/ci sub-block is wanted:
// for the convs this framework meets (Ci == Co, Cog == Cig <= 16, i.e. the
// depthwise and groups=2 stage convs of SeisT — reference models/seist.py
// uses grouped convs throughout its stages) the valid (co, ci) pairs live on
// the 16x16 diagonal tiles, and one v_mfma_f32_16x16x32_bf16 issue computes
// all 256 pair-sums of a tile for one 32-wide lo chunk. The off-group
// entries of D are simply not written back.
//
// Layout per block (256 threads, 4 waves):
//   blockIdx.x: 16-channel diagonal tile; blockIdx.y: split of the flat
//   (n, lo-chunk-of-64) work list. Each wave owns every 4th chunk and keeps
//   K f32x4 accumulators (one D fragment per tap, K <= 16 -> <= 64 VGPRs).
//   A fragment = dy[16ch x 32lo] read directly (b128, always aligned);
//   x is staged through LDS once per chunk so the K tap-shifted B fragments
//   come from LDS at arbitrary 2-byte offsets (a direct global bf16x8 load
//   at an odd tap shift would be misaligned).
//   Fragment maps as in pw_mfma.hip: A[i=lane&15][k=(lane>>4)*8+j],
//   B[k=(lane>>4)*8+j][n=lane&15], D col=lane&15, row=(lane>>4)*4+reg.
//
// Blocks write per-tile partials to a global slab (plain coalesced stores);
// a tiny second kernel reduces over the split axis. (A single-stage version
// with atomicAdd puts ~3M fp32 atomics on 26 cache lines per call — the
// two-stage form keeps the epilogue off the critical path.)

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

typedef __bf16 sa_bf16;
typedef sa_bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

namespace {

// x extent per 64-lo chunk: 8 alignment slack + (KT-1) taps + S*63 + 1,
// rounded to a multiple of 8; pitch is elems+3 (odd-ish, staggers the bank
// mapping between rows)
constexpr int x_elems(int KT, int S) {
  return (8 + (KT - 1) + S * 63 + 1 + 7) & ~7;
}

// tile -> (row ch0, col ch0): tp == 0 is the diagonal mode (Cog <= 16, the
// group mask lives inside the tile); tp >= 1 enumerates the (tp x tp) tile
// pairs of each group's diagonal block (Cog = 16*tp > 16).
__device__ __forceinline__ void tile_channels(int bx, int tp, int Cog,
                                              int* ch0r, int* ch0c) {
  if (tp == 0) {
    *ch0r = *ch0c = bx * 16;
  } else {
    const int g = bx / (tp * tp);
    const int rem = bx - g * (tp * tp);
    *ch0r = g * Cog + (rem / tp) * 16;
    *ch0c = g * Cog + (rem % tp) * 16;
  }
}

template <int KT, int S>
__global__ __launch_bounds__(256)
void dw_mfma_kernel(const sa_bf16* __restrict__ dy,
                    const sa_bf16* __restrict__ x,
                    float* __restrict__ partial,  // [gridx*zsplit][KT*256]
                    int N, int C, long L, long Lo, int K, int padl,
                    long nwork, int zsplit, int tp, int Cog) {
  constexpr int kXElems = x_elems(KT, S);
  constexpr int kXPitch = kXElems + 3;
  __shared__ sa_bf16 x_s[16 * kXPitch];
  __shared__ float red[4][256];

  int ch0, ch0c;
  tile_channels(blockIdx.x, tp, Cog, &ch0, &ch0c);
  const long per = (nwork + zsplit - 1) / zsplit;
  const long w0 = (long)blockIdx.y * per;
  const long w1 = min(nwork, w0 + per);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int frag_m = lane & 15;
  const int kbase = (lane >> 4) * 8;
  const long lochunks = (Lo + 63) / 64;

  f32x4 acc[KT];
#pragma unroll
  for (int k = 0; k < KT; ++k) acc[k] = {0.f, 0.f, 0.f, 0.f};

  const int ch = ch0 + frag_m;
  const bool chok = ch < C;

  for (long it = w0; it < w1; ++it) {
    const long n = it / lochunks;
    const long lo0 = (it - n * lochunks) * 64;

    // ---- stage x[ch0..ch0+16)[s0 .. s0+kXElems) in LDS (aligned b128) ----
    const long s0 =
        (lo0 * S - padl) & ~7L;  // aligned start; lo0*S-padl-s0 in [0,8)
    __syncthreads();
    for (int idx = tid; idx < 16 * (kXElems / 8); idx += 256) {
      const int r = idx / (kXElems / 8);
      const int c8 = idx - r * (kXElems / 8);
      const long g = s0 + c8 * 8;
      const int cg = ch0c + r;
      bf16x8 v = {};
      if (cg < C) {
        const sa_bf16* xr = x + ((long)n * C + cg) * L;
        if (g >= 0 && g + 8 <= L) {
          v = *(const bf16x8*)(xr + g);
        } else {
          for (int j = 0; j < 8; ++j) {
            const long gj = g + j;
            if (gj >= 0 && gj < L) v[j] = xr[gj];
          }
        }
      }
      *(bf16x8*)(x_s + r * kXPitch + c8 * 8) = v;
    }
    __syncthreads();

    // only wave (it % 4)'s accumulators advance this chunk; all waves help
    // stage so the LDS image is complete, but the MFMA work is divided
    if ((it & 3) != wid) continue;

    // ---- A fragments: dy 16ch x (2 x 32lo), direct aligned loads ----
    bf16x8 a0 = {}, a1 = {};
    if (chok) {
      const sa_bf16* dyr = dy + ((long)n * C + ch) * Lo;
      const long l0 = lo0 + kbase;
      if (l0 + 8 <= Lo) a0 = *(const bf16x8*)(dyr + l0);
      else for (int j = 0; j < 8; ++j)
        if (l0 + j < Lo) a0[j] = dyr[l0 + j];
      const long l1 = lo0 + 32 + kbase;
      if (l1 + 8 <= Lo) a1 = *(const bf16x8*)(dyr + l1);
      else for (int j = 0; j < 8; ++j)
        if (l1 + j < Lo) a1[j] = dyr[l1 + j];
    }

    // ---- per-tap B fragments from LDS; two K-steps per tap ----
    const int base = (int)(lo0 * S - padl - s0);  // in [0, 8)
    const sa_bf16* row = x_s + frag_m * kXPitch;
#pragma unroll
    for (int k = 0; k < KT; ++k) {
      if (k < K) {
        bf16x8 b0, b1;
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          b0[j] = row[base + k + S * (kbase + j)];
          b1[j] = row[base + k + S * (32 + kbase + j)];
        }
        acc[k] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a0, b0, acc[k], 0, 0, 0);
        acc[k] =
            __builtin_amdgcn_mfma_f32_16x16x32_bf16(a1, b1, acc[k], 0, 0, 0);
      }
    }
  }

  // ---- cross-wave reduce per tap, one coalesced partial row per tap ----
  const int d_col = lane & 15;
  const int d_row0 = (lane >> 4) * 4;
  float* out = partial + ((long)blockIdx.x * zsplit + blockIdx.y) * (KT * 256);
#pragma unroll
  for (int k = 0; k < KT; ++k) {
    if (k < K) {
      __syncthreads();
#pragma unroll
      for (int r = 0; r < 4; ++r)
        red[wid][(d_row0 + r) * 16 + d_col] = acc[k][r];
      __syncthreads();
      out[k * 256 + tid] =
          red[0][tid] + red[1][tid] + red[2][tid] + red[3][tid];
    }
  }
}

// partial[(tile*zsplit + z)][k*256 + t] -> dw[co][cig][k]
// Parallel over (tile, tap, z-chunk): each block sums <= kZChunk partial
// rows (coalesced: lane t strides the z axis) and lands one atomicAdd per
// valid (co, ci) pair — at most 8 adds per output value.
constexpr int kZChunk = 32;

template <int KT>
__global__ void dw_mfma_reduce_kernel(const float* __restrict__ partial,
                                      float* __restrict__ dw, int C, int K,
                                      int Cog, int zsplit, int tp) {
  const int tile = blockIdx.x;
  const int k = blockIdx.y;  // < K
  const long z0 = (long)blockIdx.z * kZChunk;
  const long z1 = min((long)zsplit, z0 + kZChunk);
  const int t = threadIdx.x;
  const int row = t >> 4, col = t & 15;
  int ch0r, ch0c;
  tile_channels(tile, tp, Cog, &ch0r, &ch0c);
  const int co = ch0r + row;
  const int ci = ch0c + col;
  // valid pair: same group (in diagonal mode ch0 is a multiple of Cog
  // because Cog divides 16; in tile-pair mode the pairs are in-group by
  // construction and this check is always true)
  const bool ok = co < C && ci < C && (co / Cog) == (ci / Cog);
  if (!ok) return;
  const float* base =
      partial + (long)tile * zsplit * (KT * 256) + k * 256 + t;
  float v = 0.f;
  for (long z = z0; z < z1; ++z) v += base[z * (KT * 256)];
  atomicAdd(&dw[((long)co * Cog + (ci % Cog)) * K + k], v);
}

}  // namespace

// Returns an fp32 (Co, Cig, K) weight gradient, or nullopt if the shape is
// outside this kernel's envelope (caller falls back to the direct kernel).
c10::optional<at::Tensor> dw_mfma_try(const at::Tensor& dy,
                                      const at::Tensor& x, long stride,
                                      long padl, long groups, long dilation,
                                      int K) {
  const int N = x.size(0), Ci = x.size(1);
  const long L = x.size(2);
  const int Co = dy.size(1);
  const long Lo = dy.size(2);
  const int Cig = Ci / (int)groups;
  const int Cog = Co / (int)groups;
  if (x.scalar_type() != at::kBFloat16 || dy.scalar_type() != at::kBFloat16)
    return c10::nullopt;
  if ((stride != 1 && stride != 2) || dilation != 1 || Ci != Co ||
      Cog != Cig || K > 24 || K < 1)
    return c10::nullopt;
  if (Cog > 16 ? (Cog % 16 != 0) : (16 % Cog != 0)) return c10::nullopt;

  const int C = Ci;
  const int tp = (Cog > 16) ? Cog / 16 : 0;  // 0 = diagonal-tile mode
  const int gridx = tp ? (int)groups * tp * tp : (C + 15) / 16;
  const long lochunks = (Lo + 63) / 64;
  const long nwork = (long)N * lochunks;
  // enough blocks to fill the chip on big calls, but never more splits than
  // one chunk per wave (small calls shrink the partial slab instead)
  const int zsplit = (int)std::min<long>(
      std::max<long>(1, nwork / 4),
      std::max<long>(1, 1024 / gridx));

  auto stream = at::hip::getCurrentHIPStream();
  const int KT = (K <= 8) ? 8 : (K <= 16) ? 16 : 24;
  auto partial = at::empty({(long)gridx * zsplit, (long)KT * 256},
                           x.options().dtype(at::kFloat));
  auto dw = at::zeros({(long)Co, (long)Cig, (long)K},
                      x.options().dtype(at::kFloat));

  dim3 grid(gridx, zsplit);
  auto launch = [&](auto kt, auto st) {
    hipLaunchKernelGGL(
        (dw_mfma_kernel<decltype(kt)::value, decltype(st)::value>), grid,
        dim3(256), 0, stream.stream(), (const sa_bf16*)dy.data_ptr(),
        (const sa_bf16*)x.data_ptr(), partial.data_ptr<float>(), N, C, L, Lo,
        K, (int)padl, nwork, zsplit, tp, Cog);
    dim3 rgrid(gridx, K, (zsplit + kZChunk - 1) / kZChunk);
    hipLaunchKernelGGL((dw_mfma_reduce_kernel<decltype(kt)::value>), rgrid,
                       dim3(256), 0, stream.stream(),
                       partial.data_ptr<float>(), dw.data_ptr<float>(), C, K,
                       Cog, zsplit, tp);
  };
  auto launch_s = [&](auto kt) {
    if (stride == 1) launch(kt, std::integral_constant<int, 1>{});
    else launch(kt, std::integral_constant<int, 2>{});
  };
  if (KT == 8) launch_s(std::integral_constant<int, 8>{});
  else if (KT == 16) launch_s(std::integral_constant<int, 16>{});
  else launch_s(std::integral_constant<int, 24>{});
  return dw;
}
