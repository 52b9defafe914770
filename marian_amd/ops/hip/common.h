// Shared helpers for marian_amd CDNA4 (gfx950) kernels.
// Wave size is 64 on CDNA (guide: hard-code 64, never 32).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define WAVE 64

namespace mamd {

// ---- dtype conversion -----------------------------------------------------
template <typename T> __device__ __forceinline__ float to_f32(T x);
template <> __device__ __forceinline__ float to_f32<float>(float x) { return x; }
template <> __device__ __forceinline__ float to_f32<__hip_bfloat16>(__hip_bfloat16 x) {
  return __bfloat162float(x);
}
template <> __device__ __forceinline__ float to_f32<__half>(__half x) {
  return __half2float(x);
}

template <typename T> __device__ __forceinline__ T from_f32(float x);
template <> __device__ __forceinline__ float from_f32<float>(float x) { return x; }
template <> __device__ __forceinline__ __hip_bfloat16 from_f32<__hip_bfloat16>(float x) {
  return __float2bfloat16(x);
}
template <> __device__ __forceinline__ __half from_f32<__half>(float x) {
  return __float2half(x);
}

// ---- vectorized 4-element access (8B/lane for bf16/f16, 16B for f32) -----
// hipcc does not auto-vectorize bf16 scalar loads (guide G13); these struct
// copies compile to global_load_dwordx2/x4.
template <typename T>
struct alignas(4 * sizeof(T)) Pack4 {
  T v[4];
};

template <typename T>
__device__ __forceinline__ void load4_f32(const T* __restrict__ p,
                                          float o[4]) {
  Pack4<T> t = *reinterpret_cast<const Pack4<T>*>(p);
#pragma unroll
  for (int j = 0; j < 4; ++j) o[j] = to_f32(t.v[j]);
}

template <typename T>
__device__ __forceinline__ void store4_f32(T* __restrict__ p,
                                           const float o[4]) {
  Pack4<T> t;
#pragma unroll
  for (int j = 0; j < 4; ++j) t.v[j] = from_f32<T>(o[j]);
  *reinterpret_cast<Pack4<T>*>(p) = t;
}

// ---- wave reductions (64-wide) -------------------------------------------
__device__ __forceinline__ float wave_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, WAVE);
  return v;
}

__device__ __forceinline__ float wave_max(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v = fmaxf(v, __shfl_xor(v, off, WAVE));
  return v;
}

// block reduction across waves via LDS; call with all threads.
// smem must hold blockDim.x / WAVE floats. Result valid in every thread.
__device__ __forceinline__ float block_sum(float v, float* smem) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_sum(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  const int nw = blockDim.x / WAVE;
  float r = ((int)threadIdx.x < nw) ? smem[threadIdx.x] : 0.f;
  r = wave_sum(r);  // only first wave holds all partials, but every wave
                    // reduces the same nw values read via broadcast below
  if (threadIdx.x == 0) smem[0] = r;
  __syncthreads();
  r = smem[0];
  __syncthreads();
  return r;
}

__device__ __forceinline__ float block_max(float v, float* smem) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  v = wave_max(v);
  if (lane == 0) smem[wid] = v;
  __syncthreads();
  const int nw = blockDim.x / WAVE;
  float r = ((int)threadIdx.x < nw) ? smem[threadIdx.x] : -INFINITY;
  r = wave_max(r);
  if (threadIdx.x == 0) smem[0] = r;
  __syncthreads();
  r = smem[0];
  __syncthreads();
  return r;
}

}  // namespace mamd
