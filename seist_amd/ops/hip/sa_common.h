// Common helpers for the seist_amd CDNA4 (gfx950) kernel library.
// Wave size on CDNA is 64; blocks are multiples of 64. All accumulation is
// fp32 regardless of storage dtype (bf16/fp32).
#pragma once

#include <hip/hip_runtime.h>
#include <cmath>

#define SA_CHECK_HIP(expr)                                                    \
  do {                                                                        \
    hipError_t _e = (expr);                                                   \
    if (_e != hipSuccess) {                                                   \
      TORCH_CHECK(false, "HIP error: ", hipGetErrorString(_e));               \
    }                                                                         \
  } while (0)

namespace sa {

constexpr int kWave = 64;

__device__ __forceinline__ float warp_reduce_sum(float v) {
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1) {
    v += __shfl_down(v, off, kWave);
  }
  return v;
}

// Block-wide sum reduction (block size <= 1024, multiple of 64).
// `tmp` must hold >= blockDim.x / kWave floats.
__device__ __forceinline__ float block_reduce_sum(float v, float* tmp) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wid = threadIdx.x / kWave;
  v = warp_reduce_sum(v);
  if (lane == 0) tmp[wid] = v;
  __syncthreads();
  const int nwaves = blockDim.x / kWave;
  v = (threadIdx.x < nwaves) ? tmp[threadIdx.x] : 0.0f;
  if (wid == 0) {
    v = warp_reduce_sum(v);
  }
  return v;  // valid in wave 0 (all lanes)
}

// erf-based GELU (PyTorch nn.GELU default) and its derivative.
__device__ __forceinline__ float gelu_fwd(float x) {
  return 0.5f * x * (1.0f + erff(x * 0.70710678118654752440f));
}

__device__ __forceinline__ float gelu_grad(float x) {
  const float cdf = 0.5f * (1.0f + erff(x * 0.70710678118654752440f));
  const float pdf = expf(-0.5f * x * x) * 0.39894228040143267794f;
  return cdf + x * pdf;
}

enum ActKind : int { ACT_NONE = 0, ACT_GELU = 1, ACT_RELU = 2 };

__device__ __forceinline__ float act_fwd(float x, int act) {
  if (act == ACT_GELU) return gelu_fwd(x);
  if (act == ACT_RELU) return x > 0.0f ? x : 0.0f;
  return x;
}

__device__ __forceinline__ float act_grad(float x, int act) {
  if (act == ACT_GELU) return gelu_grad(x);
  if (act == ACT_RELU) return x > 0.0f ? 1.0f : 0.0f;
  return 1.0f;
}

inline int ceil_div(long a, long b) { return (int)((a + b - 1) / b); }

}  // namespace sa
