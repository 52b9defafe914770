// Fused fp8 (OCP e4m3fn) activation quantization for CDNA4:
// ONE pass reads bf16, scales, saturate-converts via the gfx950
// v_cvt_pk_fp8_f32 instruction, writes fp8 bytes, AND reduces the
// batch amax for the delayed-scaling history (the separate
// aminmax+mul+clamp+cast chain cost 3-4 extra HBM passes and was why the
// first fp8 attempt lost to bf16 end-to-end — see docs/PERFORMANCE.md).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

using namespace mamd;

namespace {

__device__ __forceinline__ void atomic_max_f32pos(float* addr, float v) {
  // positive floats compare like their bit patterns
  atomicMax(reinterpret_cast<int*>(addr), __float_as_int(v));
}

__global__ void fp8_quantize_kernel(const __hip_bfloat16* __restrict__ x,
                                    const float* __restrict__ scale,
                                    uint8_t* __restrict__ out,
                                    float* __restrict__ amax_out,
                                    long n) {
  const float inv = 1.0f / fmaxf(*scale, 1e-12f);
  // v_cvt_pk_fp8_f32 does NOT saturate on overflow (NaN): clamp first
  auto q = [&](float v) { return fminf(fmaxf(v * inv, -448.f), 448.f); };
  float amax = 0.f;
  const long i0 = ((long)blockIdx.x * blockDim.x + threadIdx.x) * 8;
  const long stride = (long)gridDim.x * blockDim.x * 8;
  for (long i = i0; i < n; i += stride) {
    if (i + 8 <= n) {
      float v[8];
      load4_f32(x + i, v);
      load4_f32(x + i + 4, v + 4);
      uint32_t packed[2];
#pragma unroll
      for (int h = 0; h < 2; ++h) {
        const float a0 = v[4 * h], a1 = v[4 * h + 1];
        const float a2 = v[4 * h + 2], a3 = v[4 * h + 3];
        amax = fmaxf(amax, fmaxf(fmaxf(fabsf(a0), fabsf(a1)),
                                 fmaxf(fabsf(a2), fabsf(a3))));
        // cvt_pk writes 2 saturated e4m3 bytes into the selected half
        uint32_t w = __builtin_amdgcn_cvt_pk_fp8_f32(q(a0), q(a1),
                                                     0u, false);
        w = __builtin_amdgcn_cvt_pk_fp8_f32(q(a2), q(a3), w, true);
        packed[h] = w;
      }
      *reinterpret_cast<uint2*>(out + i) = make_uint2(packed[0], packed[1]);
    } else {
      for (long k = i; k < n; ++k) {
        const float a = to_f32(x[k]);
        amax = fmaxf(amax, fabsf(a));
        uint32_t w = __builtin_amdgcn_cvt_pk_fp8_f32(q(a), 0.f, 0, false);
        out[k] = (uint8_t)(w & 0xff);
      }
    }
  }
  amax = wave_max(amax);
  __shared__ float red[8];
  const int wid = threadIdx.x / 64;
  if ((threadIdx.x & 63) == 0) red[wid] = amax;
  __syncthreads();
  if (threadIdx.x == 0) {
    float m = 0.f;
    for (int w = 0; w < (int)(blockDim.x / 64); ++w) m = fmaxf(m, red[w]);
    atomic_max_f32pos(amax_out, m);
  }
}

}  // namespace

// Quantize x (bf16) to e4m3fn with the given scale (device scalar);
// returns (x8, amax) where amax is this tensor's |x| max (for the
// delayed-scaling history).
std::vector<torch::Tensor> fp8_quantize(torch::Tensor x,
                                        torch::Tensor scale) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() &&
              x.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(scale.is_cuda() && scale.scalar_type() == torch::kFloat32);
  const long n = x.numel();
  auto out = torch::empty_like(x, x.options().dtype(torch::kFloat8_e4m3fn));
  auto amax = torch::zeros({1}, x.options().dtype(torch::kFloat32));
  const int block = 256;
  const int grid = (int)std::min<long>(2048, (n + block * 8 - 1) / (block * 8));
  auto stream = at::cuda::getCurrentHIPStream();
  hipLaunchKernelGGL(fp8_quantize_kernel, dim3(grid), dim3(block), 0, stream,
                     reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                     scale.data_ptr<float>(),
                     reinterpret_cast<uint8_t*>(out.data_ptr()),
                     amax.data_ptr<float>(), n);
  return {out, amax};
}
