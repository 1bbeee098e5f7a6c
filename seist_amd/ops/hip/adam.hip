// Multi-tensor fused Adam/AdamW — K17 of SURVEY.md §2.4. The SeisT zoo
// has hundreds of tiny parameter tensors (<=1.1 M params total; the
// reference's eager torch.optim.Adam, train.py:302-323, costs several
// kernels per tensor per step). Here the whole update is ONE kernel
// launch over pre-packed chunk metadata that lives on the device and is
// built once — parameter/grad/moment pointers are stable across steps
// (grads are aliased into flat buffers by parallel.ddp.FlatReplica).
// fp32 master weights drive bf16 parameters.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "sa_common.h"

namespace {

constexpr int kBlock = 256;
constexpr int kChunk = 16384;  // elements per block

struct AdamChunk {
  void* p;
  void* g;
  float* m;
  float* v;
  float* master;  // nullptr -> update p (fp32) directly
  long off;
  long end;
};

template <typename scalar_t, bool HAS_MASTER, bool ADAMW>
__global__ void adam_kernel(const AdamChunk* __restrict__ chunks,
                            int nchunks, float lr, float beta1, float beta2,
                            float eps, float wd, float bc1, float bc2) {
  const int ci = blockIdx.x;
  if (ci >= nchunks) return;
  const AdamChunk ck = chunks[ci];

  scalar_t* p = (scalar_t*)ck.p;
  const scalar_t* g = (const scalar_t*)ck.g;

  for (long i = ck.off + threadIdx.x; i < ck.end; i += kBlock) {
    float w = HAS_MASTER ? ck.master[i] : (float)p[i];
    float gi = (float)g[i];
    if (wd != 0.0f) {
      if (ADAMW) {
        w *= (1.0f - lr * wd);
      } else {
        gi += wd * w;
      }
    }
    const float mi = ck.m[i] = beta1 * ck.m[i] + (1.0f - beta1) * gi;
    const float vi = ck.v[i] = beta2 * ck.v[i] + (1.0f - beta2) * gi * gi;
    const float denom = sqrtf(vi / bc2) + eps;
    w -= (lr / bc1) * (mi / denom);
    if (HAS_MASTER) ck.master[i] = w;
    p[i] = (scalar_t)w;
  }
}

}  // namespace

// Build device-resident chunk metadata (one blocking copy, done once).
at::Tensor adam_pack(std::vector<at::Tensor> params,
                     std::vector<at::Tensor> grads,
                     std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
                     std::vector<at::Tensor> masters, bool has_master) {
  TORCH_CHECK(!params.empty());
  std::vector<AdamChunk> chunks;
  for (size_t t = 0; t < params.size(); ++t) {
    TORCH_CHECK(params[t].is_cuda() && params[t].is_contiguous());
    TORCH_CHECK(grads[t].is_contiguous() || grads[t].numel() == 0);
    const long numel = params[t].numel();
    for (long off = 0; off < numel; off += kChunk) {
      AdamChunk ck;
      ck.p = params[t].data_ptr();
      ck.g = grads[t].data_ptr();
      ck.m = ms[t].data_ptr<float>();
      ck.v = vs[t].data_ptr<float>();
      ck.master = has_master ? masters[t].data_ptr<float>() : nullptr;
      ck.off = off;
      ck.end = std::min(numel, off + (long)kChunk);
      chunks.push_back(ck);
    }
  }
  auto meta = at::empty({(long)(sizeof(AdamChunk) * chunks.size())},
                        params[0].options().dtype(at::kByte));
  SA_CHECK_HIP(hipMemcpy(meta.data_ptr(), chunks.data(),
                         sizeof(AdamChunk) * chunks.size(),
                         hipMemcpyHostToDevice));
  return meta;
}

void adam_step_packed(const at::Tensor& meta, const at::Tensor& sample,
                      bool has_master, double lr, double beta1, double beta2,
                      double eps, double wd, double bc1, double bc2,
                      bool adamw) {
  const int nchunks = (int)(meta.numel() / sizeof(AdamChunk));
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::ScalarType::BFloat16, at::ScalarType::Half, sample.scalar_type(),
      "adam_step", [&] {
        auto launch = [&](auto has_m, auto is_w) {
          hipLaunchKernelGGL(
              (adam_kernel<scalar_t, decltype(has_m)::value,
                           decltype(is_w)::value>),
              dim3(nchunks), dim3(kBlock), 0, stream.stream(),
              (const AdamChunk*)meta.data_ptr(), nchunks, (float)lr,
              (float)beta1, (float)beta2, (float)eps, (float)wd, (float)bc1,
              (float)bc2);
        };
        if (has_master && adamw) launch(std::true_type{}, std::true_type{});
        else if (has_master) launch(std::true_type{}, std::false_type{});
        else if (adamw) launch(std::false_type{}, std::true_type{});
        else launch(std::false_type{}, std::false_type{});
      });
}
