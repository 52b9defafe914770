// LayerNorm forward + backward for CDNA4 (gfx950).
// ref semantics: src/tensors/gpu/tensor_operators.cu gLNormalization /
// gLayerNormalizationGrad — rebuilt MI355X-first: one row per 64-lane wave,
// fp32 math, vectorized 8B/lane loads (guide G13: scalar bf16 is 2-2.5x
// slower), dgamma/dbeta accumulated in registers across a wave's rows then
// one atomicAdd pass (memory-bound op: target HBM BW).
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

using namespace mamd;

namespace {

// HAS_RES: y = LN(x + res), with the sum also written to sum_out (the
// backward normalizes against the summed input) - fuses the marian "dan"
// postprocess's residual add into the LN pass.
template <typename T, int VEC, bool HAS_RES>
__global__ void ln_fwd_kernel(const T* __restrict__ x,
                              const T* __restrict__ res,
                              const T* __restrict__ gamma,
                              const T* __restrict__ beta,
                              T* __restrict__ y,
                              T* __restrict__ sum_out,
                              float* __restrict__ mean_out,
                              float* __restrict__ rstd_out,
                              int rows, int d, float eps) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  const int row = blockIdx.x * waves_per_block + wid;
  if (row >= rows) return;
  const T* xr = x + (size_t)row * d;
  const T* rr = HAS_RES ? res + (size_t)row * d : nullptr;
  T* yr = y + (size_t)row * d;
  T* sr = HAS_RES ? sum_out + (size_t)row * d : nullptr;

  float s = 0.f, ss = 0.f;
  for (int c = lane * VEC; c < d; c += WAVE * VEC) {
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float v = to_f32(xr[c + j]);
      if (HAS_RES) {
        v += to_f32(rr[c + j]);
        sr[c + j] = from_f32<T>(v);
      }
      s += v;
      ss += v * v;
    }
  }
  s = wave_sum(s);
  ss = wave_sum(ss);
  const float mu = s / d;
  const float var = fmaxf(ss / d - mu * mu, 0.f);
  const float rstd = rsqrtf(var + eps);

  // second pass re-reads the (L1/L2-resident) inputs - a register cache
  // would be runtime-indexed and spill to scratch (guide rule 20)
  for (int c = lane * VEC; c < d; c += WAVE * VEC) {
#pragma unroll
    for (int j = 0; j < VEC; ++j) {
      float v = to_f32(xr[c + j]);
      if (HAS_RES) v = to_f32(sr[c + j]);
      v = (v - mu) * rstd;
      yr[c + j] = from_f32<T>(v * to_f32(gamma[c + j]) + to_f32(beta[c + j]));
    }
  }
  if (lane == 0) {
    mean_out[row] = mu;
    rstd_out[row] = rstd;
  }
}

// backward: each WAVE walks a strided set of rows, keeping per-lane
// dgamma/dbeta partials for its columns in registers (d <= MAXD); partials
// are written per-wave to a scratch buffer (NO atomics - atomicAdd on the
// 2*d dgamma/dbeta words was 34% of a training step) and column-summed by
// ln_bwd_reduce_kernel.
template <typename T, int VEC, int MAXC>
__global__ void ln_bwd_kernel(const T* __restrict__ x,
                              const T* __restrict__ gamma,
                              const float* __restrict__ mean,
                              const float* __restrict__ rstd,
                              const T* __restrict__ dy,
                              T* __restrict__ dx,
                              float* __restrict__ dg_scratch,  // [n_waves][d]
                              float* __restrict__ db_scratch,  // [n_waves][d]
                              int rows, int d) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  const int waves_per_block = blockDim.x / WAVE;
  const int wave_id = blockIdx.x * waves_per_block + wid;
  const int n_waves = gridDim.x * waves_per_block;

  float dg[MAXC], db[MAXC];
#pragma unroll
  for (int i = 0; i < MAXC; ++i) dg[i] = db[i] = 0.f;

  for (int row = wave_id; row < rows; row += n_waves) {
    const T* xr = x + (size_t)row * d;
    const T* dyr = dy + (size_t)row * d;
    T* dxr = dx + (size_t)row * d;
    const float mu = mean[row], rs = rstd[row];

    float h1 = 0.f, h2 = 0.f;
    if constexpr (MAXC <= 16) {
      // register-cache xhat and dy*gamma across the two passes (d <= 1024
      // per 64-lane wave) - the re-read variant ran at 1.3 TB/s
      float xh[MAXC], gg[MAXC];
#pragma unroll
      for (int i = 0; i < MAXC / VEC; ++i) {
        const int c = lane * VEC + i * WAVE * VEC;
        if (c < d) {
          float x4[VEC], g4[VEC], gm4[VEC];
          load4_f32(xr + c, x4);
          load4_f32(dyr + c, g4);
          load4_f32(gamma + c, gm4);
#pragma unroll
          for (int j = 0; j < VEC; ++j) {
            const float xhv = (x4[j] - mu) * rs;
            const float ggv = g4[j] * gm4[j];
            xh[i * VEC + j] = xhv;
            gg[i * VEC + j] = ggv;
            h1 += ggv;
            h2 += ggv * xhv;
            dg[i * VEC + j] += g4[j] * xhv;
            db[i * VEC + j] += g4[j];
          }
        }
      }
      h1 = wave_sum(h1) / d;
      h2 = wave_sum(h2) / d;
#pragma unroll
      for (int i = 0; i < MAXC / VEC; ++i) {
        const int c = lane * VEC + i * WAVE * VEC;
        if (c < d) {
          float o4[VEC];
#pragma unroll
          for (int j = 0; j < VEC; ++j)
            o4[j] = rs * (gg[i * VEC + j] - h1 - xh[i * VEC + j] * h2);
          store4_f32(dxr + c, o4);
        }
      }
    } else {
      for (int c = lane * VEC, i = 0; c < d; c += WAVE * VEC) {
#pragma unroll
        for (int j = 0; j < VEC; ++j, ++i) {
          const float xh = (to_f32(xr[c + j]) - mu) * rs;
          const float g = to_f32(dyr[c + j]);
          const float gg = g * to_f32(gamma[c + j]);
          h1 += gg;
          h2 += gg * xh;
          dg[i] += g * xh;
          db[i] += g;
        }
      }
      h1 = wave_sum(h1) / d;
      h2 = wave_sum(h2) / d;
      for (int c = lane * VEC; c < d; c += WAVE * VEC) {
#pragma unroll
        for (int j = 0; j < VEC; ++j) {
          const float xh = (to_f32(xr[c + j]) - mu) * rs;
          const float gg = to_f32(dyr[c + j]) * to_f32(gamma[c + j]);
          dxr[c + j] = from_f32<T>(rs * (gg - h1 - xh * h2));
        }
      }
    }
  }
  float* dgrow = dg_scratch + (size_t)wave_id * d;
  float* dbrow = db_scratch + (size_t)wave_id * d;
  for (int c = lane * VEC, i = 0; c < d; c += WAVE * VEC) {
#pragma unroll
    for (int j = 0; j < VEC; ++j, ++i) {
      dgrow[c + j] = dg[i];
      dbrow[c + j] = db[i];
    }
  }
}

// column-sum the per-wave partials: dgamma[c] = sum_w scratch[w][c].
// 2-D grid: blockIdx.x tiles columns, blockIdx.y tiles row stripes, so the
// sum is latency-parallel; stripes combine with a handful of atomics.
__global__ void ln_bwd_reduce_kernel(const float* __restrict__ dg_scratch,
                                     const float* __restrict__ db_scratch,
                                     float* __restrict__ dgamma,
                                     float* __restrict__ dbeta,
                                     int n_waves, int d) {
  const int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= d) return;
  const int stripes = gridDim.y;
  const int per = (n_waves + stripes - 1) / stripes;
  const int w0 = blockIdx.y * per;
  const int w1 = min(w0 + per, n_waves);
  float sg = 0.f, sb = 0.f;
  for (int w = w0; w < w1; ++w) {
    sg += dg_scratch[(size_t)w * d + c];
    sb += db_scratch[(size_t)w * d + c];
  }
  if (stripes == 1) {
    dgamma[c] = sg;
    dbeta[c] = sb;
  } else {
    atomicAdd(&dgamma[c], sg);
    atomicAdd(&dbeta[c], sb);
  }
}

constexpr int kBlock = 256;

template <typename T>
void ln_fwd_launch(const torch::Tensor& x, const torch::Tensor* res,
                   const torch::Tensor& g,
                   const torch::Tensor& b, torch::Tensor& y,
                   torch::Tensor* sum_out,
                   torch::Tensor& mean, torch::Tensor& rstd, double eps) {
  const int d = x.size(-1);
  const int rows = x.numel() / d;
  TORCH_CHECK(d % 4 == 0, "layer_norm: dim must be divisible by 4, got ", d);
  const int waves = kBlock / WAVE;
  const int grid = (rows + waves - 1) / waves;
  auto stream = at::cuda::getCurrentHIPStream();
  if (res) {
    hipLaunchKernelGGL((ln_fwd_kernel<T, 4, true>), dim3(grid), dim3(kBlock),
                       0, stream,
                       reinterpret_cast<const T*>(x.data_ptr()),
                       reinterpret_cast<const T*>(res->data_ptr()),
                       reinterpret_cast<const T*>(g.data_ptr()),
                       reinterpret_cast<const T*>(b.data_ptr()),
                       reinterpret_cast<T*>(y.data_ptr()),
                       reinterpret_cast<T*>(sum_out->data_ptr()),
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       rows, d, (float)eps);
  } else {
    hipLaunchKernelGGL((ln_fwd_kernel<T, 4, false>), dim3(grid), dim3(kBlock),
                       0, stream,
                       reinterpret_cast<const T*>(x.data_ptr()), nullptr,
                       reinterpret_cast<const T*>(g.data_ptr()),
                       reinterpret_cast<const T*>(b.data_ptr()),
                       reinterpret_cast<T*>(y.data_ptr()), nullptr,
                       mean.data_ptr<float>(), rstd.data_ptr<float>(),
                       rows, d, (float)eps);
  }
}

template <typename T>
void ln_bwd_launch(const torch::Tensor& x, const torch::Tensor& g,
                   const torch::Tensor& mean, const torch::Tensor& rstd,
                   const torch::Tensor& dy, torch::Tensor& dx,
                   torch::Tensor& dgamma, torch::Tensor& dbeta) {
  const int d = x.size(-1);
  const int rows = x.numel() / d;
  TORCH_CHECK(d % 4 == 0 && d <= 64 * 64,
              "layer_norm bwd: dim must be /4 and <= 4096, got ", d);
  const int waves = kBlock / WAVE;
  // enough waves to fill the chip, few enough that the scratch reduce
  // stays cheap (2048 blocks x 4 waves -> 8192 partial rows max)
  int grid = std::min((rows + waves - 1) / waves, 1024);
  const int n_waves = grid * waves;
  auto opts = x.options().dtype(torch::kFloat32);
  auto dg_scratch = torch::empty({n_waves, d}, opts);
  auto db_scratch = torch::empty({n_waves, d}, opts);
  auto stream = at::cuda::getCurrentHIPStream();
#define LAUNCH_BWD(MAXC)                                                     \
  hipLaunchKernelGGL((ln_bwd_kernel<T, 4, MAXC>), dim3(grid), dim3(kBlock),  \
                     0, stream, reinterpret_cast<const T*>(x.data_ptr()),    \
                     reinterpret_cast<const T*>(g.data_ptr()),               \
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),         \
                     reinterpret_cast<const T*>(dy.data_ptr()),              \
                     reinterpret_cast<T*>(dx.data_ptr()),                    \
                     dg_scratch.data_ptr<float>(),                           \
                     db_scratch.data_ptr<float>(), rows, d)
  const int per_lane = (d + WAVE - 1) / WAVE;  // columns per lane
  if (per_lane <= 8) {
    LAUNCH_BWD(8);
  } else if (per_lane <= 16) {
    LAUNCH_BWD(16);
  } else if (per_lane <= 32) {
    LAUNCH_BWD(32);
  } else {
    LAUNCH_BWD(64);
  }
#undef LAUNCH_BWD
  // waves with no rows still write their (zero) register partials, so the
  // whole scratch is defined.
  const int stripes = std::max(1, std::min(n_waves / 64, 64));
  hipLaunchKernelGGL(ln_bwd_reduce_kernel,
                     dim3((d + 255) / 256, stripes), dim3(256),
                     0, stream, dg_scratch.data_ptr<float>(),
                     db_scratch.data_ptr<float>(), dgamma.data_ptr<float>(),
                     dbeta.data_ptr<float>(), n_waves, d);
}

}  // namespace

// returns (y, mean, rstd)
std::vector<torch::Tensor> layernorm_fwd(torch::Tensor x, torch::Tensor gamma,
                                         torch::Tensor beta, double eps) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  auto y = torch::empty_like(x);
  const int d = x.size(-1);
  const int rows = x.numel() / d;
  auto opts = x.options().dtype(torch::kFloat32);
  auto mean = torch::empty({rows}, opts);
  auto rstd = torch::empty({rows}, opts);
  auto g = gamma.contiguous(), b = beta.contiguous();
  if (x.scalar_type() == torch::kBFloat16) {
    ln_fwd_launch<__hip_bfloat16>(x, nullptr, g, b, y, nullptr, mean, rstd, eps);
  } else if (x.scalar_type() == torch::kHalf) {
    ln_fwd_launch<__half>(x, nullptr, g, b, y, nullptr, mean, rstd, eps);
  } else if (x.scalar_type() == torch::kFloat32) {
    ln_fwd_launch<float>(x, nullptr, g, b, y, nullptr, mean, rstd, eps);
  } else {
    TORCH_CHECK(false, "layernorm: unsupported dtype");
  }
  return {y, mean, rstd};
}

// y = LN(x + res); returns (y, sum, mean, rstd)
std::vector<torch::Tensor> layernorm_res_fwd(torch::Tensor x,
                                             torch::Tensor res,
                                             torch::Tensor gamma,
                                             torch::Tensor beta, double eps) {
  TORCH_CHECK(x.is_cuda());
  auto xc = x.contiguous();
  auto rc = res.contiguous();
  auto y = torch::empty_like(xc);
  auto sum = torch::empty_like(xc);
  const int d = xc.size(-1);
  const int rows = xc.numel() / d;
  auto opts = xc.options().dtype(torch::kFloat32);
  auto mean = torch::empty({rows}, opts);
  auto rstd = torch::empty({rows}, opts);
  auto g = gamma.contiguous(), b = beta.contiguous();
  if (xc.scalar_type() == torch::kBFloat16) {
    ln_fwd_launch<__hip_bfloat16>(xc, &rc, g, b, y, &sum, mean, rstd, eps);
  } else if (xc.scalar_type() == torch::kHalf) {
    ln_fwd_launch<__half>(xc, &rc, g, b, y, &sum, mean, rstd, eps);
  } else if (xc.scalar_type() == torch::kFloat32) {
    ln_fwd_launch<float>(xc, &rc, g, b, y, &sum, mean, rstd, eps);
  } else {
    TORCH_CHECK(false, "layernorm: unsupported dtype");
  }
  return {y, sum, mean, rstd};
}

// returns (dx, dgamma_f32, dbeta_f32)
std::vector<torch::Tensor> layernorm_bwd(torch::Tensor x, torch::Tensor gamma,
                                         torch::Tensor mean, torch::Tensor rstd,
                                         torch::Tensor dy) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  auto dyc = dy.contiguous();
  auto dx = torch::empty_like(x);
  const int d = x.size(-1);
  auto opts = x.options().dtype(torch::kFloat32);
  auto dgamma = torch::zeros({d}, opts);
  auto dbeta = torch::zeros({d}, opts);
  auto g = gamma.contiguous();
  if (x.scalar_type() == torch::kBFloat16) {
    ln_bwd_launch<__hip_bfloat16>(x, g, mean, rstd, dyc, dx, dgamma, dbeta);
  } else if (x.scalar_type() == torch::kHalf) {
    ln_bwd_launch<__half>(x, g, mean, rstd, dyc, dx, dgamma, dbeta);
  } else if (x.scalar_type() == torch::kFloat32) {
    ln_bwd_launch<float>(x, g, mean, rstd, dyc, dx, dgamma, dbeta);
  } else {
    TORCH_CHECK(false, "layernorm: unsupported dtype");
  }
  return {dx, dgamma, dbeta};
}
