// Fused label-smoothed cross-entropy for CDNA4 (gfx950).
// ref semantics: src/tensors/gpu/tensor_operators.cu gCrossEntropyPick /
// gCrossEntropyPickBackward + layers/loss.cpp label smoothing.
//
// Forward: one block per row (vocab up to 32k+), ONE pass over the logits
// with online (max, sumexp, sum) accumulation; emits per-row loss
//   (1-eps) * (lse - x[label]) + eps * (lse - mean(x))
// and saves lse for the backward.
// Backward writes softmax(x) - (1-eps)*onehot - eps/V scaled by the
// incoming per-row gradient — no softmax materialization in between.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

using namespace mamd;

namespace {

constexpr int kBlock = 256;

template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              const float* __restrict__ mask,
                              float* __restrict__ loss,
                              float* __restrict__ lse_out,
                              int rows, int cols, float eps) {
  __shared__ float smem[kBlock / WAVE];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = logits + (size_t)row * cols;
    const int64_t label = labels[row];
    float m = -INFINITY, s = 0.f, xsum = 0.f, xlab = 0.f;
    const int cols4 = (cols % 4 == 0) ? cols : 0;  // unaligned rows -> scalar path
    for (int c = threadIdx.x * 4; c < cols4; c += blockDim.x * 4) {
      float v4[4];
      load4_f32(xr + c, v4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float v = v4[j];
        xsum += v;
        if (c + j == (int)label) xlab = v;
        if (v > m) {
          s = s * __expf(m - v) + 1.f;
          m = v;
        } else {
          s += __expf(v - m);
        }
      }
    }
    for (int c = cols4 + threadIdx.x; c < cols; c += blockDim.x) {
      const float v = to_f32(xr[c]);
      xsum += v;
      if (c == (int)label) xlab = v;
      if (v > m) {
        s = s * __expf(m - v) + 1.f;
        m = v;
      } else {
        s += __expf(v - m);
      }
    }
    const float gm = block_max(m, smem);
    float s_scaled = (m == -INFINITY) ? 0.f : s * __expf(m - gm);
    const float gs = block_sum(s_scaled, smem);
    const float lse = gm + __logf(gs);
    const float gxsum = block_sum(xsum, smem);
    const float gxlab = block_sum(xlab, smem);  // only the owner added it
    if (threadIdx.x == 0) {
      const float w = mask ? mask[row] : 1.f;
      float l = (1.f - eps) * (lse - gxlab) + eps * (lse - gxsum / cols);
      loss[row] = l * w;
      lse_out[row] = lse;
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void ce_bwd_kernel(const T* __restrict__ logits,
                              const int64_t* __restrict__ labels,
                              const float* __restrict__ mask,
                              const float* __restrict__ lse,
                              const float* __restrict__ dloss,
                              T* __restrict__ dx,
                              int rows, int cols, float eps) {
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = logits + (size_t)row * cols;
    T* dxr = dx + (size_t)row * cols;
    const int64_t label = labels[row];
    const float l = lse[row];
    const float w = (mask ? mask[row] : 1.f) * dloss[row];
    const float smooth = eps / cols;
    const int cols4 = (cols % 4 == 0) ? cols : 0;  // unaligned rows -> scalar path
    for (int c = threadIdx.x * 4; c < cols4; c += blockDim.x * 4) {
      float v4[4];
      load4_f32(xr + c, v4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        float p = __expf(v4[j] - l);
        p -= (c + j == (int)label) ? (1.f - eps) : 0.f;
        v4[j] = (p - smooth) * w;
      }
      store4_f32(dxr + c, v4);
    }
    for (int c = cols4 + threadIdx.x; c < cols; c += blockDim.x) {
      float p = __expf(to_f32(xr[c]) - l);
      p -= (c == (int)label) ? (1.f - eps) : 0.f;
      p -= smooth;
      dxr[c] = from_f32<T>((p)*w);
    }
  }
}

}  // namespace

// returns (loss_f32 [rows], lse_f32 [rows])
std::vector<torch::Tensor> ce_fwd(torch::Tensor logits, torch::Tensor labels,
                                  double eps,
                                  c10::optional<torch::Tensor> mask) {
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 2);
  auto x = logits.contiguous();
  auto lab = labels.contiguous();
  TORCH_CHECK(lab.scalar_type() == torch::kInt64);
  const int rows = x.size(0), cols = x.size(1);
  auto opts = x.options().dtype(torch::kFloat32);
  auto loss = torch::empty({rows}, opts);
  auto lse = torch::empty({rows}, opts);
  torch::Tensor maskc;
  const float* mask_ptr = nullptr;
  if (mask.has_value() && mask->defined()) {
    maskc = mask->to(torch::kFloat32).contiguous();
    mask_ptr = maskc.data_ptr<float>();
  }
  const int grid = std::min(rows, 8192);
  auto stream = at::cuda::getCurrentHIPStream();
#define RUN(T)                                                               \
  hipLaunchKernelGGL((ce_fwd_kernel<T>), dim3(grid), dim3(kBlock), 0,        \
                     stream, reinterpret_cast<const T*>(x.data_ptr()),       \
                     lab.data_ptr<int64_t>(), mask_ptr,                      \
                     loss.data_ptr<float>(), lse.data_ptr<float>(), rows,    \
                     cols, (float)eps)
  if (x.scalar_type() == torch::kBFloat16) RUN(__hip_bfloat16);
  else if (x.scalar_type() == torch::kHalf) RUN(__half);
  else if (x.scalar_type() == torch::kFloat32) RUN(float);
  else TORCH_CHECK(false, "ce: unsupported dtype");
#undef RUN
  return {loss, lse};
}

torch::Tensor ce_bwd(torch::Tensor logits, torch::Tensor labels,
                     torch::Tensor lse, torch::Tensor dloss, double eps,
                     c10::optional<torch::Tensor> mask) {
  auto x = logits.contiguous();
  auto lab = labels.contiguous();
  auto dl = dloss.to(torch::kFloat32).contiguous();
  const int rows = x.size(0), cols = x.size(1);
  auto dx = torch::empty_like(x);
  torch::Tensor maskc;
  const float* mask_ptr = nullptr;
  if (mask.has_value() && mask->defined()) {
    maskc = mask->to(torch::kFloat32).contiguous();
    mask_ptr = maskc.data_ptr<float>();
  }
  const int grid = std::min(rows, 8192);
  auto stream = at::cuda::getCurrentHIPStream();
#define RUN(T)                                                               \
  hipLaunchKernelGGL((ce_bwd_kernel<T>), dim3(grid), dim3(kBlock), 0,        \
                     stream, reinterpret_cast<const T*>(x.data_ptr()),       \
                     lab.data_ptr<int64_t>(), mask_ptr,                      \
                     lse.data_ptr<float>(), dl.data_ptr<float>(),            \
                     reinterpret_cast<T*>(dx.data_ptr()), rows, cols,        \
                     (float)eps)
  if (x.scalar_type() == torch::kBFloat16) RUN(__hip_bfloat16);
  else if (x.scalar_type() == torch::kHalf) RUN(__half);
  else if (x.scalar_type() == torch::kFloat32) RUN(float);
  else TORCH_CHECK(false, "ce: unsupported dtype");
#undef RUN
  return dx;
}
