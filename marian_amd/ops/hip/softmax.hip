// Row softmax / log-softmax forward + backward for CDNA4 (gfx950).
// ref semantics: src/tensors/gpu/tensor_operators.cu gSoftmax/gLogSoftmax
// (+Grad). Block-per-row with online (m,s) reduction, fp32 math, bf16 I/O.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

using namespace mamd;

namespace {

constexpr int kBlock = 256;

template <typename T, bool LOG>
__global__ void softmax_fwd_kernel(const T* __restrict__ x,
                                   T* __restrict__ y,
                                   int rows, int cols) {
  __shared__ float smem[kBlock / WAVE];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (size_t)row * cols;
    T* yr = y + (size_t)row * cols;
    // online max+sum in one pass over the row (8B/lane vectorized)
    float m = -INFINITY, s = 0.f;
    const int cols4 = (cols % 4 == 0) ? cols : 0;  // unaligned rows -> scalar path
    for (int c = threadIdx.x * 4; c < cols4; c += blockDim.x * 4) {
      float v4[4];
      load4_f32(xr + c, v4);
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const float v = v4[j];
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
      if (v > m) {
        s = s * __expf(m - v) + 1.f;
        m = v;
      } else {
        s += __expf(v - m);
      }
    }
    // combine (m, s) across the block: reduce max, then rescaled sums
    const float gm = block_max(m, smem);
    float s_scaled = (m == -INFINITY) ? 0.f : s * __expf(m - gm);
    const float gs = block_sum(s_scaled, smem);
    const float inv = 1.f / gs;
    const float lse = gm + __logf(gs);
    for (int c = threadIdx.x * 4; c < cols4; c += blockDim.x * 4) {
      float v4[4];
      load4_f32(xr + c, v4);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        v4[j] = LOG ? (v4[j] - lse) : __expf(v4[j] - gm) * inv;
      store4_f32(yr + c, v4);
    }
    for (int c = cols4 + threadIdx.x; c < cols; c += blockDim.x) {
      const float v = to_f32(xr[c]);
      yr[c] = from_f32<T>(LOG ? (v - lse) : __expf(v - gm) * inv);
    }
  }
}

template <typename T, bool LOG>
__global__ void softmax_bwd_kernel(const T* __restrict__ y,
                                   const T* __restrict__ dy,
                                   T* __restrict__ dx,
                                   int rows, int cols) {
  __shared__ float smem[kBlock / WAVE];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* yr = y + (size_t)row * cols;
    const T* dyr = dy + (size_t)row * cols;
    T* dxr = dx + (size_t)row * cols;
    float acc = 0.f;
    const int cols4 = (cols % 4 == 0) ? cols : 0;  // unaligned rows -> scalar path
    for (int c = threadIdx.x * 4; c < cols4; c += blockDim.x * 4) {
      float g4[4], y4[4];
      load4_f32(dyr + c, g4);
      load4_f32(yr + c, y4);
#pragma unroll
      for (int j = 0; j < 4; ++j) acc += LOG ? g4[j] : g4[j] * y4[j];
    }
    for (int c = cols4 + threadIdx.x; c < cols; c += blockDim.x) {
      const float g = to_f32(dyr[c]);
      acc += LOG ? g : g * to_f32(yr[c]);
    }
    const float tot = block_sum(acc, smem);
    // softmax: dx = (dy - sum(dy*y)) * y ; logsoftmax: dx = dy - exp(y)*sum(dy)
    for (int c = threadIdx.x * 4; c < cols4; c += blockDim.x * 4) {
      float g4[4], y4[4];
      load4_f32(dyr + c, g4);
      load4_f32(yr + c, y4);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        g4[j] = LOG ? (g4[j] - __expf(y4[j]) * tot) : (g4[j] - tot) * y4[j];
      store4_f32(dxr + c, g4);
    }
    for (int c = cols4 + threadIdx.x; c < cols; c += blockDim.x) {
      const float g = to_f32(dyr[c]);
      const float yv = to_f32(yr[c]);
      dxr[c] = from_f32<T>(LOG ? (g - __expf(yv) * tot) : (g - tot) * yv);
    }
  }
}

template <bool LOG>
torch::Tensor softmax_fwd_impl(torch::Tensor x) {
  TORCH_CHECK(x.is_cuda());
  auto xc = x.contiguous();
  auto y = torch::empty_like(xc);
  const int cols = xc.size(-1);
  const int rows = xc.numel() / cols;
  const int grid = std::min(rows, 8192);
  auto stream = at::cuda::getCurrentHIPStream();
#define RUN(T)                                                            \
  hipLaunchKernelGGL((softmax_fwd_kernel<T, LOG>), dim3(grid),            \
                     dim3(kBlock), 0, stream,                             \
                     reinterpret_cast<const T*>(xc.data_ptr()),           \
                     reinterpret_cast<T*>(y.data_ptr()), rows, cols)
  if (xc.scalar_type() == torch::kBFloat16) RUN(__hip_bfloat16);
  else if (xc.scalar_type() == torch::kHalf) RUN(__half);
  else if (xc.scalar_type() == torch::kFloat32) RUN(float);
  else TORCH_CHECK(false, "softmax: unsupported dtype");
#undef RUN
  return y;
}

template <bool LOG>
torch::Tensor softmax_bwd_impl(torch::Tensor y, torch::Tensor dy) {
  auto yc = y.contiguous();
  auto dyc = dy.contiguous();
  auto dx = torch::empty_like(yc);
  const int cols = yc.size(-1);
  const int rows = yc.numel() / cols;
  const int grid = std::min(rows, 8192);
  auto stream = at::cuda::getCurrentHIPStream();
#define RUN(T)                                                            \
  hipLaunchKernelGGL((softmax_bwd_kernel<T, LOG>), dim3(grid),            \
                     dim3(kBlock), 0, stream,                             \
                     reinterpret_cast<const T*>(yc.data_ptr()),           \
                     reinterpret_cast<const T*>(dyc.data_ptr()),          \
                     reinterpret_cast<T*>(dx.data_ptr()), rows, cols)
  if (yc.scalar_type() == torch::kBFloat16) RUN(__hip_bfloat16);
  else if (yc.scalar_type() == torch::kHalf) RUN(__half);
  else if (yc.scalar_type() == torch::kFloat32) RUN(float);
  else TORCH_CHECK(false, "softmax: unsupported dtype");
#undef RUN
  return dx;
}

}  // namespace

torch::Tensor softmax_fwd(torch::Tensor x) { return softmax_fwd_impl<false>(x); }
torch::Tensor log_softmax_fwd(torch::Tensor x) { return softmax_fwd_impl<true>(x); }
torch::Tensor softmax_bwd(torch::Tensor y, torch::Tensor dy) {
  return softmax_bwd_impl<false>(y, dy);
}
torch::Tensor log_softmax_bwd(torch::Tensor y, torch::Tensor dy) {
  return softmax_bwd_impl<true>(y, dy);
}
