// Fused Adam step for CDNA4 (gfx950).
// ref semantics: src/optimizers/optimizers.cpp Adam::updateImpl (the
// Element(...) fused functor). One elementwise pass over the ZeRO-1 shard:
// fp32 master/state, bf16/fp32 gradient, bias-corrected update.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

using namespace mamd;

namespace {

template <typename GT>
__global__ void adam_kernel(float* __restrict__ p,
                            const GT* __restrict__ g,
                            float* __restrict__ m,
                            float* __restrict__ v,
                            long n, float lr, float b1, float b2, float eps,
                            float wd, float bc1, float bc2, float gscale,
                            const float* __restrict__ gscale_dev,
                            const float* __restrict__ lr_dev,
                            const int* __restrict__ step_dev) {
  // device-resident scale factor (cost-type/clip computed on GPU so the
  // host never syncs mid-step); uniform scalar load, hits L2 once per CU.
  // lr_dev/step_dev: hipGraph-replay mode — the step counter and LR live
  // on device so bias correction stays correct across graph replays.
  if (gscale_dev) gscale *= gscale_dev[0];
  if (lr_dev) lr = lr_dev[0];
  if (step_dev) {
    const float tt = (float)step_dev[0];
    bc1 = 1.f - powf(b1, tt);
    bc2 = 1.f - powf(b2, tt);
  }
  const long i0 = (long)(blockIdx.x) * blockDim.x + threadIdx.x;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = i0; i < n; i += stride) {
    const float gi = to_f32(g[i]) * gscale;
    float mi = b1 * m[i] + (1.f - b1) * gi;
    float vi = b2 * v[i] + (1.f - b2) * gi * gi;
    m[i] = mi;
    v[i] = vi;
    const float mh = mi / bc1;
    const float vh = vi / bc2;
    float pi = p[i];
    if (wd != 0.f) pi *= (1.f - lr * wd);
    p[i] = pi - lr * mh / (sqrtf(vh) + eps);
  }
}

}  // namespace

void adam_step(torch::Tensor p, torch::Tensor g, torch::Tensor m,
               torch::Tensor v, double lr, double b1, double b2, double eps,
               double wd, int64_t t, double gscale,
               c10::optional<torch::Tensor> gscale_t,
               c10::optional<torch::Tensor> lr_t,
               c10::optional<torch::Tensor> step_t) {
  TORCH_CHECK(p.is_cuda() && p.scalar_type() == torch::kFloat32);
  TORCH_CHECK(p.is_contiguous() && m.is_contiguous() && v.is_contiguous());
  const float* gsp = nullptr;
  if (gscale_t.has_value()) {
    TORCH_CHECK(gscale_t->is_cuda() &&
                gscale_t->scalar_type() == torch::kFloat32 &&
                gscale_t->is_contiguous());
    gsp = gscale_t->data_ptr<float>();
  }
  const float* lrp = nullptr;
  if (lr_t.has_value()) {
    TORCH_CHECK(lr_t->is_cuda() && lr_t->scalar_type() == torch::kFloat32);
    lrp = lr_t->data_ptr<float>();
  }
  const int* stp = nullptr;
  if (step_t.has_value()) {
    TORCH_CHECK(step_t->is_cuda() && step_t->scalar_type() == torch::kInt32);
    stp = step_t->data_ptr<int>();
  }
  auto gc = g.contiguous();
  const long n = p.numel();
  const int block = 256;
  const int grid = (int)std::min<long>((n + block - 1) / block, 4096);
  const float bc1 = 1.f - powf((float)b1, (float)t);
  const float bc2 = 1.f - powf((float)b2, (float)t);
  auto stream = at::cuda::getCurrentHIPStream();
#define RUN(GT)                                                              \
  hipLaunchKernelGGL((adam_kernel<GT>), dim3(grid), dim3(block), 0, stream,  \
                     p.data_ptr<float>(),                                    \
                     reinterpret_cast<const GT*>(gc.data_ptr()),             \
                     m.data_ptr<float>(), v.data_ptr<float>(), n, (float)lr, \
                     (float)b1, (float)b2, (float)eps, (float)wd, bc1, bc2,  \
                     (float)gscale, gsp, lrp, stp)
  if (gc.scalar_type() == torch::kBFloat16) RUN(__hip_bfloat16);
  else if (gc.scalar_type() == torch::kHalf) RUN(__half);
  else if (gc.scalar_type() == torch::kFloat32) RUN(float);
  else TORCH_CHECK(false, "adam: unsupported grad dtype");
#undef RUN
}
