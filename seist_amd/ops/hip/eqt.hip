// EQTransformer-family kernels — K8 (LayerNorm), K10 (additive/banded
// attention) and K11 (LSTM recurrence) of SURVEY §2.4.
//
// Replaces the reference's library stages (models/eqtransformer.py:105-198,
// 245-262 and models/magnet.py:95-101):
//  * additive attention: score e_ij = Wa·tanh(q_i + k_j + bh) + ba with
//    exp/band-mask/sum-eps normalisation — fused so the (N,L,L,d) tanh
//    tensor (262 MB at N=500) never exists; only the (N,L,L) weights do.
//  * LayerNorm over the channel-last (N*L, C) rows.
//  * LSTM: input projections are one big GEMM outside; the sequential
//    recurrence runs as a persistent kernel (wave-group per sample,
//    W_hh staged in LDS, two in-block barriers per timestep) instead of
//    MIOpen's per-step launches.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

namespace {

constexpr int kBlock = 256;

// ---------------------------------------------------------------------------
// K8: LayerNorm over rows of C (channel-last), fp32 stats
// ---------------------------------------------------------------------------

template <typename scalar_t>
__global__ void ln_fwd_kernel(const scalar_t* __restrict__ x,
                              scalar_t* __restrict__ y,
                              const float* __restrict__ gamma,
                              const float* __restrict__ beta,
                              float* __restrict__ mean_out,
                              float* __restrict__ rstd_out,
                              long nrows, int C, float eps) {
  const int lane = threadIdx.x & (sa::kWave - 1);
  const long row = (long)blockIdx.x * (blockDim.x / sa::kWave)
                   + threadIdx.x / sa::kWave;
  if (row >= nrows) return;
  const scalar_t* xr = x + row * C;
  float s = 0.0f, s2 = 0.0f;
  for (int c = lane; c < C; c += sa::kWave) {
    const float v = (float)xr[c];
    s += v;
    s2 += v * v;
  }
  s = sa::warp_reduce_sum(s);
  s2 = sa::warp_reduce_sum(s2);
  s = __shfl(s, 0, sa::kWave);
  s2 = __shfl(s2, 0, sa::kWave);
  const float m = s / C;
  const float var = fmaxf(s2 / C - m * m, 0.0f);
  const float rstd = rsqrtf(var + eps);
  if (lane == 0) {
    mean_out[row] = m;
    rstd_out[row] = rstd;
  }
  scalar_t* yr = y + row * C;
  for (int c = lane; c < C; c += sa::kWave) {
    yr[c] = (scalar_t)(((float)xr[c] - m) * rstd * gamma[c] + beta[c]);
  }
}

template <typename scalar_t>
__global__ void ln_bwd_kernel(const scalar_t* __restrict__ dy,
                              const scalar_t* __restrict__ x,
                              scalar_t* __restrict__ dx,
                              const float* __restrict__ gamma,
                              const float* __restrict__ mean,
                              const float* __restrict__ rstd,
                              float* __restrict__ dgb_part,  // (nblk, 2C)
                              long nrows, int C) {
  extern __shared__ float lds[];  // 2C accumulators for dgamma/dbeta
  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) lds[i] = 0.0f;
  __syncthreads();
  const int lane = threadIdx.x & (sa::kWave - 1);
  const int rpb = blockDim.x / sa::kWave;
  const long row = (long)blockIdx.x * rpb + threadIdx.x / sa::kWave;
  if (row < nrows) {
    const scalar_t* xr = x + row * C;
    const scalar_t* dyr = dy + row * C;
    const float m = mean[row], rs = rstd[row];
    float sg = 0.0f, sgx = 0.0f;
    for (int c = lane; c < C; c += sa::kWave) {
      const float xh = ((float)xr[c] - m) * rs;
      const float g = (float)dyr[c] * gamma[c];
      sg += g;
      sgx += g * xh;
      // per-channel partials (atomic into LDS: C is small, rows per block
      // contend only when C < 64)
      atomicAdd(&lds[c], (float)dyr[c]);
      atomicAdd(&lds[C + c], (float)dyr[c] * xh);
    }
    sg = sa::warp_reduce_sum(sg);
    sgx = sa::warp_reduce_sum(sgx);
    sg = __shfl(sg, 0, sa::kWave) / C;
    sgx = __shfl(sgx, 0, sa::kWave) / C;
    scalar_t* dxr = dx + row * C;
    for (int c = lane; c < C; c += sa::kWave) {
      const float xh = ((float)xr[c] - m) * rs;
      const float g = (float)dyr[c] * gamma[c];
      dxr[c] = (scalar_t)((g - sg - xh * sgx) * rs);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < 2 * C; i += blockDim.x) {
    dgb_part[(long)blockIdx.x * 2 * C + i] = lds[i];
  }
}

// ---------------------------------------------------------------------------
// K10: additive attention scores + softmax
//   e~_ij = Wa . tanh(q_i + k_j + bh) + ba
//   u = exp(e~ - rowmax);  e = band ? u : 0;  a = e / (sum e + eps)
// one wave per query row i; rows of one sample share a block so the
// k-tile (L x d) is staged once into LDS.
// ---------------------------------------------------------------------------

constexpr int kAttnRows = 4;    // query rows (waves) per block
constexpr float kAttnEps = 1e-6f;

__global__ void addattn_fwd_kernel(const float* __restrict__ q,   // (N,L,d)
                                   const float* __restrict__ k,   // (N,L,d)
                                   const float* __restrict__ bh,  // (d)
                                   const float* __restrict__ wa,  // (d)
                                   const float* __restrict__ ba_p,
                                   float* __restrict__ attn,      // (N,L,L)
                                   float* __restrict__ ssum,      // (N,L)
                                   int* __restrict__ amax,        // (N,L)
                                   int L, int d, int tril_k, int triu_k) {
  extern __shared__ float lds[];           // k tile (L*d) + q rows
  float* kt = lds;                         // L*d
  float* qt = lds + (long)L * d;           // kAttnRows*d
  const int n = blockIdx.x / (L / kAttnRows);
  const int i0 = (blockIdx.x % (L / kAttnRows)) * kAttnRows;
  const int wid = threadIdx.x / sa::kWave;
  const int lane = threadIdx.x & (sa::kWave - 1);
  const int i = i0 + wid;

  for (int t = threadIdx.x; t < L * d; t += blockDim.x) {
    kt[t] = k[((long)n * L) * d + t];
  }
  for (int t = threadIdx.x; t < kAttnRows * d; t += blockDim.x) {
    qt[t] = q[((long)n * L + i0) * d + t];
  }
  __syncthreads();

  // scores for this row, lanes strided over j
  float ev[4];                              // supports L <= 256
  const int ntile = (L + sa::kWave - 1) / sa::kWave;
  float m = -1e30f;
  int mj = 0;
#pragma unroll
  for (int tIdx = 0; tIdx < 4; ++tIdx) {
    if (tIdx >= ntile) break;
    const int j = tIdx * sa::kWave + lane;
    float e = -1e30f;
    if (j < L) {
      e = ba_p[0];
      const float* kj = kt + (long)j * d;
      const float* qi = qt + (long)wid * d;
      for (int c = 0; c < d; ++c) {
        e += wa[c] * tanhf(qi[c] + kj[c] + bh[c]);
      }
      if (e > m) { m = e; mj = j; }
    }
    ev[tIdx] = e;
  }
  // wave max + argmax (first occurrence on ties = lowest j wins strictly
  // greater comparison ordering below)
#pragma unroll
  for (int off = sa::kWave / 2; off > 0; off >>= 1) {
    const float om = __shfl_down(m, off, sa::kWave);
    const int oj = __shfl_down(mj, off, sa::kWave);
    if (om > m || (om == m && oj < mj)) { m = om; mj = oj; }
  }
  m = __shfl(m, 0, sa::kWave);
  mj = __shfl(mj, 0, sa::kWave);

  float s = 0.0f;
#pragma unroll
  for (int tIdx = 0; tIdx < 4; ++tIdx) {
    if (tIdx >= ntile) break;
    const int j = tIdx * sa::kWave + lane;
    if (j < L) {
      float u = expf(ev[tIdx] - m);
      const int dj = j - i;
      if (dj > tril_k || dj < triu_k) u = 0.0f;
      ev[tIdx] = u;
      s += u;
    }
  }
  s = sa::warp_reduce_sum(s);
  s = __shfl(s, 0, sa::kWave) + kAttnEps;

  float* ar = attn + ((long)n * L + i) * L;
#pragma unroll
  for (int tIdx = 0; tIdx < 4; ++tIdx) {
    if (tIdx >= ntile) break;
    const int j = tIdx * sa::kWave + lane;
    if (j < L) ar[j] = ev[tIdx] / s;
  }
  if (lane == 0) {
    ssum[(long)n * L + i] = s;
    amax[(long)n * L + i] = mj;
  }
}

// de~ from (a, da): g_j = a_j * (da_j - sum_k a_k da_k);
// de~_j = g_j - [j == argmax] * sum_k g_k
__global__ void addattn_descore_kernel(const float* __restrict__ attn,
                                       const float* __restrict__ dattn,
                                       const int* __restrict__ amax,
                                       float* __restrict__ descore,  // (N,L,L)
                                       int L) {
  const long row = blockIdx.x * (blockDim.x / sa::kWave)
                   + threadIdx.x / sa::kWave;
  const int lane = threadIdx.x & (sa::kWave - 1);
  const float* ar = attn + row * L;
  const float* dar = dattn + row * L;
  float* out = descore + row * L;
  const int ntile = (L + sa::kWave - 1) / sa::kWave;

  float sdot = 0.0f;
  for (int t = 0; t < ntile; ++t) {
    const int j = t * sa::kWave + lane;
    if (j < L) sdot += ar[j] * dar[j];
  }
  sdot = sa::warp_reduce_sum(sdot);
  sdot = __shfl(sdot, 0, sa::kWave);

  float gs = 0.0f;
  for (int t = 0; t < ntile; ++t) {
    const int j = t * sa::kWave + lane;
    if (j < L) {
      const float g = ar[j] * (dar[j] - sdot);
      out[j] = g;
      gs += g;
    }
  }
  gs = sa::warp_reduce_sum(gs);
  gs = __shfl(gs, 0, sa::kWave);
  if (lane == 0) out[amax[row]] -= gs;
}

// dq_i,d = sum_j de~_ij * wa_d * (1 - t_ijd^2), t recomputed; also
// accumulates per-block dWa partials (sum_ij de~ * t) and dba partials.
__global__ void addattn_dq_kernel(const float* __restrict__ q,
                                  const float* __restrict__ k,
                                  const float* __restrict__ bh,
                                  const float* __restrict__ wa,
                                  const float* __restrict__ descore,
                                  float* __restrict__ dq,       // (N,L,d)
                                  float* __restrict__ dwa_part, // (nblk, d+1)
                                  int L, int d) {
  extern __shared__ float lds[];      // k tile (L*d) + dwa accum (d+1)
  float* kt = lds;
  float* acc = lds + (long)L * d;
  const int n = blockIdx.x / (L / kAttnRows);
  const int i0 = (blockIdx.x % (L / kAttnRows)) * kAttnRows;
  const int wid = threadIdx.x / sa::kWave;
  const int lane = threadIdx.x & (sa::kWave - 1);
  const int i = i0 + wid;

  for (int t = threadIdx.x; t < L * d; t += blockDim.x) {
    kt[t] = k[((long)n * L) * d + t];
  }
  for (int t = threadIdx.x; t < d + 1; t += blockDim.x) acc[t] = 0.0f;
  __syncthreads();

  const float* qi = q + ((long)n * L + i) * d;
  const float* de = descore + ((long)n * L + i) * L;
  float* dqi = dq + ((long)n * L + i) * d;

  // lanes handle channels; loop j serial (d <= 64 assumed for lane map)
  float deba = 0.0f;
  if (lane < d) {
    const float qv = qi[lane];
    const float bv = bh[lane];
    const float wv = wa[lane];
    float acc_dq = 0.0f, acc_dwa = 0.0f;
    for (int j = 0; j < L; ++j) {
      const float t = tanhf(qv + kt[(long)j * d + lane] + bv);
      const float dej = de[j];
      acc_dq += dej * wv * (1.0f - t * t);
      acc_dwa += dej * t;
      if (lane == 0) deba += dej;
    }
    dqi[lane] = acc_dq;
    atomicAdd(&acc[lane], acc_dwa);
    if (lane == 0) atomicAdd(&acc[d], deba);
  }
  __syncthreads();
  for (int t = threadIdx.x; t < d + 1; t += blockDim.x) {
    dwa_part[(long)blockIdx.x * (d + 1) + t] = acc[t];
  }
}

// dk_j,d = sum_i de~_ij * wa_d * (1 - t_ijd^2): same shape, transposed loop
__global__ void addattn_dk_kernel(const float* __restrict__ q,
                                  const float* __restrict__ k,
                                  const float* __restrict__ bh,
                                  const float* __restrict__ wa,
                                  const float* __restrict__ descore,
                                  float* __restrict__ dk,  // (N,L,d)
                                  int L, int d) {
  extern __shared__ float lds[];  // q tile (L*d)
  float* qt = lds;
  const int n = blockIdx.x / (L / kAttnRows);
  const int j0 = (blockIdx.x % (L / kAttnRows)) * kAttnRows;
  const int wid = threadIdx.x / sa::kWave;
  const int lane = threadIdx.x & (sa::kWave - 1);
  const int j = j0 + wid;

  for (int t = threadIdx.x; t < L * d; t += blockDim.x) {
    qt[t] = q[((long)n * L) * d + t];
  }
  __syncthreads();

  const float* kj = k + ((long)n * L + j) * d;
  const float* de = descore + (long)n * L * L;
  float* dkj = dk + ((long)n * L + j) * d;
  if (lane < d) {
    const float kv = kj[lane];
    const float bv = bh[lane];
    const float wv = wa[lane];
    float acc = 0.0f;
    for (int i = 0; i < L; ++i) {
      const float t = tanhf(qt[(long)i * d + lane] + kv + bv);
      acc += de[(long)i * L + j] * wv * (1.0f - t * t);
    }
    dkj[lane] = acc;
  }
}

// ---------------------------------------------------------------------------
// K11: LSTM recurrence. Input projections (x @ W_ih^T + b_ih + b_hh) are
// one GEMM outside; this kernel runs the time loop with W_hh staged in LDS.
// Geometry: G = ceil(4H/64) waves per sample; spb samples per block.
// PyTorch gate order [i, f, g, o]; c/h state in LDS.
// ---------------------------------------------------------------------------

__global__ void lstm_fwd_kernel(const float* __restrict__ pre,  // (N,L,4H)
                                const float* __restrict__ whh,  // (4H,H)
                                float* __restrict__ y,     // (N,L,D*H)
                                float* __restrict__ cstash,  // (N,L,H)
                                float* __restrict__ gstash,  // (N,L,4H) or null
                                int N, int L, int H, int dirs, int dir,
                                int spb) {
  extern __shared__ float lds[];
  const int G4 = 4 * H;
  float* w = lds;                        // 4H*H
  float* state = lds + (long)G4 * H;     // per sample: h[H], c[H], gates[4H]
  const int gwaves = (G4 + sa::kWave - 1) / sa::kWave;
  const int slot = (threadIdx.x / sa::kWave) / gwaves;   // sample slot
  const int gid = (threadIdx.x % ((long)gwaves * sa::kWave));
  const int n = blockIdx.x * spb + slot;

  // W staged TRANSPOSED [hh][g]: the per-hh inner loop then reads
  // consecutive g across threads — conflict-free broadcast instead of a
  // 32-way bank conflict on the row-major stride-H access
  for (int t = threadIdx.x; t < G4 * H; t += blockDim.x) {
    const int g_ = t / H;
    const int hh_ = t - g_ * H;
    w[hh_ * G4 + g_] = whh[t];
  }
  float* h = state + (long)slot * (2 * H + G4);
  float* c = h + H;
  float* gates = c + H;
  for (int t = gid; t < 2 * H + G4; t += gwaves * sa::kWave) h[t] = 0.0f;
  __syncthreads();

  const bool active = (n < N) && (gid < G4);
  const int g = gid;
  for (int step = 0; step < L; ++step) {
    const int t = dir ? (L - 1 - step) : step;
    if (active) {
      float acc = pre[((long)n * L + t) * G4 + g];
      for (int hh = 0; hh < H; ++hh) acc += w[hh * G4 + g] * h[hh];
      const int kind = g / H;
      gates[g] = (kind == 2) ? tanhf(acc)
                             : 1.0f / (1.0f + expf(-acc));
    }
    __syncthreads();
    if (active && g < H) {
      const float iv = gates[g];
      const float fv = gates[H + g];
      const float gv = gates[2 * H + g];
      const float ov = gates[3 * H + g];
      const float cv = fv * c[g] + iv * gv;
      c[g] = cv;
      const float hv = ov * tanhf(cv);
      h[g] = hv;
      y[((long)n * L + t) * (dirs * H) + dir * H + g] = hv;
      if (cstash != nullptr) cstash[((long)n * L + t) * H + g] = cv;
    }
    if (active && gstash != nullptr) {
      gstash[((long)n * L + t) * G4 + g] = gates[g];
    }
    __syncthreads();
  }
}

// Wave-per-sample variant for 4H <= 64 (EQT hidden 16): the whole gate
// vector fits one wavefront, so the h/c exchange runs on cross-lane
// shuffles with ZERO barriers and W_hh lives in registers — samples never
// wait on each other (the workgroup variant's per-step __syncthreads made
// 8 independent samples stride together: 170 us/call vs ~8 us here).
__global__ void lstm_fwd_wave_kernel(const float* __restrict__ pre,
                                     const float* __restrict__ whh,
                                     float* __restrict__ y,
                                     float* __restrict__ cstash,
                                     float* __restrict__ gstash,
                                     int N, int L, int H, int dirs,
                                     int dir) {
  const int spb = blockDim.x / sa::kWave;
  const int slot = threadIdx.x >> 6;
  const int lane = threadIdx.x & (sa::kWave - 1);
  const int n = blockIdx.x * spb + slot;
  const int G4 = 4 * H;
  const bool active = (n < N) && (lane < G4);

  float w[16];
#pragma unroll
  for (int hh = 0; hh < 16; ++hh) {
    w[hh] = (active && hh < H) ? whh[(long)lane * H + hh] : 0.0f;
  }
  float hcur = 0.0f, ccur = 0.0f;  // valid on lanes < H

  for (int step = 0; step < L; ++step) {
    const int t = dir ? (L - 1 - step) : step;
    const long base = ((long)n * L + t);
    float acc = active ? pre[base * G4 + lane] : 0.0f;
    // compile-time bound so w[] stays in registers (w is zero-padded
    // past H, so the extra shuffles contribute nothing)
#pragma unroll
    for (int hh = 0; hh < 16; ++hh) {
      acc += w[hh] * __shfl(hcur, hh, sa::kWave);
    }
    const float gate = (lane / H == 2) ? tanhf(acc)
                                       : 1.0f / (1.0f + expf(-acc));
    const float iv = __shfl(gate, lane, sa::kWave);
    const float fv = __shfl(gate, H + lane, sa::kWave);
    const float gv = __shfl(gate, 2 * H + lane, sa::kWave);
    const float ov = __shfl(gate, 3 * H + lane, sa::kWave);
    if (active && lane < H) {
      ccur = fv * ccur + iv * gv;
      hcur = ov * tanhf(ccur);
      y[base * (dirs * H) + dir * H + lane] = hcur;
      if (cstash != nullptr) cstash[base * H + lane] = ccur;
    }
    if (active && gstash != nullptr) gstash[base * G4 + lane] = gate;
  }
}

__global__ void lstm_bwd_wave_kernel(const float* __restrict__ dy,
                                     const float* __restrict__ cstash,
                                     const float* __restrict__ gstash,
                                     const float* __restrict__ whh,
                                     float* __restrict__ dgates,
                                     int N, int L, int H, int dirs,
                                     int dir) {
  const int spb = blockDim.x / sa::kWave;
  const int slot = threadIdx.x >> 6;
  const int lane = threadIdx.x & (sa::kWave - 1);
  const int n = blockIdx.x * spb + slot;
  const int G4 = 4 * H;
  const bool active = (n < N) && (lane < G4);
  const int h = lane % H;
  const int chunk = lane / H;

  // W column h in registers (lanes < H need it for the dh recursion)
  float wcol[64];
#pragma unroll 16
  for (int g = 0; g < 64; ++g) {
    wcol[g] = (active && lane < H && g < G4)
        ? whh[(long)g * H + lane] : 0.0f;
  }

  float dh = 0.0f, dc = 0.0f;  // lanes < H
  for (int step = L - 1; step >= 0; --step) {
    const int t = dir ? (L - 1 - step) : step;
    const long base = ((long)n * L + t);
    const float gown = active ? gstash[base * G4 + lane] : 0.0f;
    // shuffles must run on ALL lanes (a shfl under a divergent branch
    // reads zeros from exec-masked source lanes)
    const float ov_l = __shfl(gown, 3 * H + h, sa::kWave);
    const float fv_l = __shfl(gown, H + h, sa::kWave);
    float dhv = 0.0f, dcv = 0.0f, tc = 0.0f, cprev = 0.0f;
    if (active && lane < H) {
      const float cv = cstash[base * H + lane];
      tc = tanhf(cv);
      dhv = dh + dy[base * (dirs * H) + dir * H + lane];
      dcv = dc + dhv * ov_l * (1.0f - tc * tc);
      cprev = (step == 0)
          ? 0.0f
          : cstash[((long)n * L + (dir ? (L - step) : (t - 1))) * H + lane];
    }
    // broadcast per-h quantities to all gate lanes
    const float dcv_h = __shfl(dcv, h, sa::kWave);
    const float dhv_h = __shfl(dhv, h, sa::kWave);
    const float tc_h = __shfl(tc, h, sa::kWave);
    const float cprev_h = __shfl(cprev, h, sa::kWave);
    const float i_h = __shfl(gown, h, sa::kWave);            // i-chunk value
    const float g_h = __shfl(gown, 2 * H + h, sa::kWave);    // g-chunk value
    float dg = 0.0f;
    if (chunk == 0) dg = dcv_h * g_h * gown * (1.0f - gown);
    else if (chunk == 1) dg = dcv_h * cprev_h * gown * (1.0f - gown);
    else if (chunk == 2) dg = dcv_h * i_h * (1.0f - gown * gown);
    else dg = dhv_h * tc_h * gown * (1.0f - gown);
    if (active) dgates[base * G4 + lane] = dg;
    // dh_{t-1}[h] = sum_g w[g][h] * dg[g];  dc_{t-1} = dcv * f
    // compile-time bound keeps wcol in registers (zero-padded past G4)
    float acc = 0.0f;
#pragma unroll
    for (int g = 0; g < 64; ++g) {
      acc += wcol[g] * __shfl(dg, g, sa::kWave);
    }
    if (active && lane < H) {
      dh = acc;
      dc = dcv * fv_l;
    }
  }
}

// reverse-time BPTT; consumes the stashes; emits per-element dgates
// (N,L,4H) — dW_ih/dW_hh/db and dx are GEMMs outside.
__global__ void lstm_bwd_kernel(const float* __restrict__ dy,  // (N,L,D*H)
                                const float* __restrict__ y,   // (N,L,D*H)
                                const float* __restrict__ cstash,
                                const float* __restrict__ gstash,
                                const float* __restrict__ whh,  // (4H,H)
                                float* __restrict__ dgates,     // (N,L,4H)
                                int N, int L, int H, int dirs, int dir,
                                int spb) {
  extern __shared__ float lds[];
  const int G4 = 4 * H;
  float* w = lds;                         // 4H*H
  float* state = lds + (long)G4 * H;      // per sample: dh[H], dc[H], dg[4H]
  const int gwaves = (G4 + sa::kWave - 1) / sa::kWave;
  const int slot = (threadIdx.x / sa::kWave) / gwaves;
  const int gid = (threadIdx.x % ((long)gwaves * sa::kWave));
  const int n = blockIdx.x * spb + slot;

  for (int t = threadIdx.x; t < G4 * H; t += blockDim.x) w[t] = whh[t];
  float* dh = state + (long)slot * (2 * H + G4);
  float* dc = dh + H;
  float* dg = dc + H;
  for (int t = gid; t < 2 * H + G4; t += gwaves * sa::kWave) dh[t] = 0.0f;
  __syncthreads();

  const bool active = (n < N) && (gid < G4);
  const int g = gid;
  for (int step = L - 1; step >= 0; --step) {
    const int t = dir ? (L - 1 - step) : step;
    // phase 1: threads g < H update dh/dc and compute gate grads
    if (active && g < H) {
      const long base = ((long)n * L + t);
      const float cv = cstash[base * H + g];
      const float tc = tanhf(cv);
      const float iv = gstash[base * G4 + g];
      const float fv = gstash[base * G4 + H + g];
      const float gv = gstash[base * G4 + 2 * H + g];
      const float ov = gstash[base * G4 + 3 * H + g];
      float dhv = dh[g] + dy[base * (dirs * H) + dir * H + g];
      float dcv = dc[g] + dhv * ov * (1.0f - tc * tc);
      const float cprev = (step == 0)
          ? 0.0f
          : cstash[((long)n * L + (dir ? (L - step) : (t - 1))) * H + g];
      dg[g] = dcv * gv * iv * (1.0f - iv);                // d(pre_i)
      dg[H + g] = dcv * cprev * fv * (1.0f - fv);         // d(pre_f)
      dg[2 * H + g] = dcv * iv * (1.0f - gv * gv);        // d(pre_g)
      dg[3 * H + g] = dhv * tc * ov * (1.0f - ov);        // d(pre_o)
      dc[g] = dcv * fv;
    }
    __syncthreads();
    // phase 2: all gate threads write dgates; h-threads compute dh_{t-1}
    if (active) {
      dgates[((long)n * L + t) * G4 + g] = dg[g];
    }
    __syncthreads();
    if (active && g < H) {
      // w[gg*H + g]: per gg, threads h read consecutive addresses —
      // already conflict-free in the row-major image
      float acc = 0.0f;
      for (int gg = 0; gg < G4; ++gg) acc += w[(long)gg * H + g] * dg[gg];
      dh[g] = acc;
    }
    __syncthreads();
  }
}

}  // namespace

// ---------------------------------------------------------------------------
// host entry points
// ---------------------------------------------------------------------------

std::vector<at::Tensor> ln_fwd(const at::Tensor& x, const at::Tensor& gamma,
                               const at::Tensor& beta, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int C = x.size(-1);
  const long nrows = x.numel() / C;
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = x.options().dtype(at::kFloat);
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto b32 = beta.to(at::kFloat).contiguous();
  auto y = at::empty_like(x);
  auto mean = at::empty({nrows}, opts);
  auto rstd = at::empty({nrows}, opts);
  const int rpb = kBlock / sa::kWave;
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "ln_fwd", [&] {
        hipLaunchKernelGGL((ln_fwd_kernel<scalar_t>),
                           dim3(sa::ceil_div(nrows, rpb)), dim3(kBlock), 0,
                           stream.stream(), x.data_ptr<scalar_t>(),
                           y.data_ptr<scalar_t>(), g32.data_ptr<float>(),
                           b32.data_ptr<float>(), mean.data_ptr<float>(),
                           rstd.data_ptr<float>(), nrows, C, (float)eps);
      });
  return {y, mean, rstd};
}

std::vector<at::Tensor> ln_bwd(const at::Tensor& dy, const at::Tensor& x,
                               const at::Tensor& gamma,
                               const at::Tensor& mean,
                               const at::Tensor& rstd) {
  const int C = x.size(-1);
  const long nrows = x.numel() / C;
  auto stream = at::hip::getCurrentHIPStream();
  auto g32 = gamma.to(at::kFloat).contiguous();
  auto dx = at::empty_like(x);
  const int rpb = kBlock / sa::kWave;
  const int nblk = sa::ceil_div(nrows, rpb);
  auto part = at::empty({nblk, 2 * C}, x.options().dtype(at::kFloat));
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, x.scalar_type(),
      "ln_bwd", [&] {
        hipLaunchKernelGGL((ln_bwd_kernel<scalar_t>), dim3(nblk),
                           dim3(kBlock), 2 * C * sizeof(float),
                           stream.stream(), dy.data_ptr<scalar_t>(),
                           x.data_ptr<scalar_t>(), dx.data_ptr<scalar_t>(),
                           g32.data_ptr<float>(), mean.data_ptr<float>(),
                           rstd.data_ptr<float>(), part.data_ptr<float>(),
                           nrows, C);
      });
  auto sums = part.sum(0);
  auto dbeta = sums.narrow(0, 0, C).to(gamma.scalar_type());
  auto dgamma = sums.narrow(0, C, C).to(gamma.scalar_type());
  return {dx, dgamma, dbeta};
}

std::vector<at::Tensor> addattn_fwd(const at::Tensor& q, const at::Tensor& k,
                                    const at::Tensor& bh, const at::Tensor& wa,
                                    const at::Tensor& ba, long tril_k,
                                    long triu_k) {
  TORCH_CHECK(q.is_cuda() && q.is_contiguous() && k.is_contiguous());
  TORCH_CHECK(q.scalar_type() == at::kFloat, "additive attention is fp32");
  const int N = q.size(0), L = q.size(1), d = q.size(2);
  TORCH_CHECK(L % kAttnRows == 0 && L <= 256 && d <= 64);
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = q.options();
  auto attn = at::empty({N, L, L}, opts);
  auto ssum = at::empty({N, L}, opts);
  auto amax = at::empty({N, L}, opts.dtype(at::kInt));
  const int lds = (L * d + kAttnRows * d) * sizeof(float);
  hipLaunchKernelGGL(addattn_fwd_kernel,
                     dim3((long)N * (L / kAttnRows)),
                     dim3(kAttnRows * sa::kWave), lds, stream.stream(),
                     q.data_ptr<float>(), k.data_ptr<float>(),
                     bh.data_ptr<float>(), wa.data_ptr<float>(),
                     ba.data_ptr<float>(),
                     attn.data_ptr<float>(), ssum.data_ptr<float>(),
                     amax.data_ptr<int>(), L, d, (int)tril_k, (int)triu_k);
  return {attn, ssum, amax};
}

std::vector<at::Tensor> addattn_bwd(const at::Tensor& q, const at::Tensor& k,
                                    const at::Tensor& bh, const at::Tensor& wa,
                                    const at::Tensor& attn,
                                    const at::Tensor& dattn,
                                    const at::Tensor& amax) {
  const int N = q.size(0), L = q.size(1), d = q.size(2);
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = q.options();
  auto descore = at::empty({N, L, L}, opts);
  auto dattn_c = dattn.contiguous();
  const int rpb = kBlock / sa::kWave;
  hipLaunchKernelGGL(addattn_descore_kernel,
                     dim3(sa::ceil_div((long)N * L, rpb)), dim3(kBlock), 0,
                     stream.stream(), attn.data_ptr<float>(),
                     dattn_c.data_ptr<float>(),
                     amax.data_ptr<int>(), descore.data_ptr<float>(), L);
  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  const long nblk = (long)N * (L / kAttnRows);
  auto dwa_part = at::empty({nblk, d + 1}, opts);
  hipLaunchKernelGGL(addattn_dq_kernel, dim3(nblk),
                     dim3(kAttnRows * sa::kWave),
                     (L * d + d + 1) * sizeof(float), stream.stream(),
                     q.data_ptr<float>(), k.data_ptr<float>(),
                     bh.data_ptr<float>(), wa.data_ptr<float>(),
                     descore.data_ptr<float>(), dq.data_ptr<float>(),
                     dwa_part.data_ptr<float>(), L, d);
  hipLaunchKernelGGL(addattn_dk_kernel, dim3(nblk),
                     dim3(kAttnRows * sa::kWave), L * d * sizeof(float),
                     stream.stream(), q.data_ptr<float>(),
                     k.data_ptr<float>(), bh.data_ptr<float>(),
                     wa.data_ptr<float>(), descore.data_ptr<float>(),
                     dk.data_ptr<float>(), L, d);
  auto sums = dwa_part.sum(0);
  auto dwa = sums.narrow(0, 0, d);
  auto dba = sums.narrow(0, d, 1);
  return {dq, dk, dwa, dba, descore};
}

std::vector<at::Tensor> lstm_fwd(const at::Tensor& pre, const at::Tensor& whh,
                                 at::Tensor y, long dir, bool training) {
  TORCH_CHECK(pre.is_cuda() && pre.is_contiguous() && whh.is_contiguous());
  TORCH_CHECK(pre.scalar_type() == at::kFloat && y.is_contiguous());
  const int N = pre.size(0), L = pre.size(1);
  const int H = pre.size(2) / 4;
  const int dirs = y.size(2) / H;
  auto stream = at::hip::getCurrentHIPStream();
  auto opts = pre.options();
  const int G4 = 4 * H;
  const int gwaves = sa::ceil_div(G4, sa::kWave);
  // samples per block: keep blocks <= 512 threads
  const int spb = std::max(1, 512 / (gwaves * sa::kWave));
  const long slds = (long)spb * (2 * H + G4) * sizeof(float);
  const long lds = (long)G4 * H * sizeof(float) + slds;
  TORCH_CHECK(lds <= 160 * 1024, "LSTM hidden too large for LDS staging");
  auto cstash = training ? at::empty({N, L, H}, opts) : at::Tensor();
  auto gstash = training ? at::empty({N, L, G4}, opts) : at::Tensor();
  if (G4 <= sa::kWave) {
    // wave-per-sample, barrier-free (H <= 16)
    const int wspb = 4;
    hipLaunchKernelGGL(lstm_fwd_wave_kernel,
                       dim3(sa::ceil_div(N, wspb)),
                       dim3(wspb * sa::kWave), 0, stream.stream(),
                       pre.data_ptr<float>(), whh.data_ptr<float>(),
                       y.data_ptr<float>(),
                       training ? cstash.data_ptr<float>() : nullptr,
                       training ? gstash.data_ptr<float>() : nullptr,
                       N, L, H, dirs, (int)dir);
  } else {
    hipLaunchKernelGGL(lstm_fwd_kernel, dim3(sa::ceil_div(N, spb)),
                       dim3(spb * gwaves * sa::kWave), lds, stream.stream(),
                       pre.data_ptr<float>(), whh.data_ptr<float>(),
                       y.data_ptr<float>(),
                       training ? cstash.data_ptr<float>() : nullptr,
                       training ? gstash.data_ptr<float>() : nullptr,
                       N, L, H, dirs, (int)dir, spb);
  }
  if (training) return {cstash, gstash};
  return {};
}

at::Tensor lstm_bwd(const at::Tensor& dy, const at::Tensor& y,
                    const at::Tensor& cstash, const at::Tensor& gstash,
                    const at::Tensor& whh, long dirs, long dir) {
  const int N = dy.size(0), L = dy.size(1);
  const int H = cstash.size(2);
  const int G4 = 4 * H;
  auto stream = at::hip::getCurrentHIPStream();
  const int gwaves = sa::ceil_div(G4, sa::kWave);
  const int spb = std::max(1, 512 / (gwaves * sa::kWave));
  const long lds = (long)G4 * H * sizeof(float)
                   + (long)spb * (2 * H + G4) * sizeof(float);
  auto dgates = at::empty({N, L, G4}, dy.options());
  auto dy_c = dy.contiguous();
  if (G4 <= sa::kWave) {
    const int wspb = 4;
    hipLaunchKernelGGL(lstm_bwd_wave_kernel, dim3(sa::ceil_div(N, wspb)),
                       dim3(wspb * sa::kWave), 0, stream.stream(),
                       dy_c.data_ptr<float>(), cstash.data_ptr<float>(),
                       gstash.data_ptr<float>(), whh.data_ptr<float>(),
                       dgates.data_ptr<float>(), N, L, H, (int)dirs,
                       (int)dir);
  } else {
    hipLaunchKernelGGL(lstm_bwd_kernel, dim3(sa::ceil_div(N, spb)),
                       dim3(spb * gwaves * sa::kWave), lds, stream.stream(),
                       dy_c.data_ptr<float>(), y.data_ptr<float>(),
                       cstash.data_ptr<float>(), gstash.data_ptr<float>(),
                       whh.data_ptr<float>(), dgates.data_ptr<float>(),
                       N, L, H, (int)dirs, (int)dir, spb);
  }
  return dgates;
}
