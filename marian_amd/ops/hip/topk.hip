// Fused beam-search top-k for CDNA4 (gfx950).
// ref: src/translator/nth_element.cu (NthElementGPU gMaxElement /
// gMaxElementUpdate) + helpers.cu gSetColumn — the reference's batched
// top-k over beam*vocab with forbidden-token suppression.
//
// One block per sentence. Computes top-k of (logprobs[b*K+kb][v] +
// path_scores[b*K+kb]) over the flattened beam*vocab candidate space
// WITHOUT materializing the candidate tensor: each thread keeps a sorted
// top-k of its strided slice in registers, then a two-stage LDS merge.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

using namespace mamd;

namespace {

constexpr int kBlock = 256;
constexpr int MAXK = 16;
constexpr float NEG_BIGF = -1e30f;

struct Cand {
  float v;
  int i;
};

template <int K>
__device__ __forceinline__ void insert(Cand (&top)[K], float v, int idx) {
  if (v <= top[K - 1].v) return;
  int p = K - 1;
  while (p > 0 && top[p - 1].v < v) {
    top[p] = top[p - 1];
    --p;
  }
  top[p].v = v;
  top[p].i = idx;
}

template <typename T, int K>
__global__ void beam_topk_kernel(const T* __restrict__ logprobs,  // [B*NB, V]
                                 const float* __restrict__ scores,  // [B*NB]
                                 float* __restrict__ out_v,  // [B, k_out]
                                 long* __restrict__ out_i,   // [B, k_out]
                                 int B, int NB, long V, int suppress,
                                 int k_out) {
  __shared__ Cand smem[kBlock / WAVE * K];
  const int b = blockIdx.x;
  const int tid = threadIdx.x;
  const int lane = tid & (WAVE - 1);
  const int wid = tid / WAVE;

  Cand top[K];
#pragma unroll
  for (int i = 0; i < K; ++i) top[i] = {NEG_BIGF, -1};

  for (int kb = 0; kb < NB; ++kb) {
    const T* row = logprobs + ((size_t)b * NB + kb) * V;
    const float s = scores[b * NB + kb];
    if (s <= -1e8f) continue;  // dead beam (python NEG_INF = -1e9)
    const long base = (long)kb * V;
    const long V4 = (V % 4 == 0) ? V : 0;
    for (long c = (long)tid * 4; c < V4; c += (long)blockDim.x * 4) {
      float v4[4];
      load4_f32(row + c, v4);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        if (c + j != suppress) insert(top, v4[j] + s, (int)(base + c + j));
    }
    for (long c = V4 + tid; c < V; c += blockDim.x) {
      if (c != suppress) insert(top, to_f32(row[c]) + s, (int)(base + c));
    }
  }

  // wave-level merge: lane l shares entry via shuffle-based reduction.
  // Simpler two-stage: each lane dumps its K to a wave-collective pick by
  // iterated argmax over the wave (K iterations of a wave max-reduce).
  Cand wave_top[K];
  int my_pos = 0;  // next unconsumed entry of this lane's sorted list
#pragma unroll
  for (int k = 0; k < K; ++k) {
    float cand_v = (my_pos < K) ? top[my_pos].v : NEG_BIGF;
    int cand_i = (my_pos < K) ? top[my_pos].i : -1;
    // wave argmax of cand_v
    float best_v = cand_v;
    int best_l = lane;
#pragma unroll
    for (int off = 32; off > 0; off >>= 1) {
      float ov = __shfl_xor(best_v, off, WAVE);
      int ol = __shfl_xor(best_l, off, WAVE);
      if (ov > best_v || (ov == best_v && ol < best_l)) {
        best_v = ov;
        best_l = ol;
      }
    }
    const int src_idx = __shfl(cand_i, best_l, WAVE);
    wave_top[k] = {best_v, src_idx};
    if (lane == best_l) ++my_pos;
  }
  if (lane == 0) {
#pragma unroll
    for (int k = 0; k < K; ++k) smem[wid * K + k] = wave_top[k];
  }
  __syncthreads();
  // final merge of (kBlock/WAVE) sorted lists by thread 0..K-1's wave
  if (wid == 0) {
    int pos[kBlock / WAVE];
#pragma unroll
    for (int w = 0; w < kBlock / WAVE; ++w) pos[w] = 0;
    if (lane == 0) {
      for (int k = 0; k < k_out; ++k) {
        float best_v = NEG_BIGF;
        int best_w = -1;
#pragma unroll
        for (int w = 0; w < kBlock / WAVE; ++w) {
          if (pos[w] < K) {
            const float v = smem[w * K + pos[w]].v;
            if (v > best_v) {
              best_v = v;
              best_w = w;
            }
          }
        }
        if (best_w < 0) {
          out_v[(size_t)b * k_out + k] = NEG_BIGF;
          out_i[(size_t)b * k_out + k] = 0;
        } else {
          out_v[(size_t)b * k_out + k] = smem[best_w * K + pos[best_w]].v;
          out_i[(size_t)b * k_out + k] = smem[best_w * K + pos[best_w]].i;
          ++pos[best_w];
        }
      }
    }
  }
}

}  // namespace

// logprobs: [B*NB, V]; scores: [B*NB] f32 path scores; returns
// (values [B,K] f32, indices [B,K] long into the flattened NB*V space).
std::vector<torch::Tensor> beam_topk(torch::Tensor logprobs,
                                     torch::Tensor scores, int64_t n_beams,
                                     int64_t k, int64_t suppress) {
  TORCH_CHECK(logprobs.is_cuda() && logprobs.dim() == 2);
  TORCH_CHECK(k >= 1 && k <= MAXK, "beam_topk: k must be in [1,16]");
  auto lp = logprobs.contiguous();
  auto sc = scores.to(torch::kFloat32).contiguous();
  const long rows = lp.size(0);
  const long V = lp.size(1);
  const int NB = (int)n_beams;
  TORCH_CHECK(rows % NB == 0);
  const int B = (int)(rows / NB);
  TORCH_CHECK((long)NB * V < (1LL << 31), "beam*vocab must fit int32");
  auto out_v = torch::empty({B, k}, lp.options().dtype(torch::kFloat32));
  auto out_i = torch::empty({B, k}, lp.options().dtype(torch::kInt64));
  auto stream = at::cuda::getCurrentHIPStream();
#define RUNK(T, KV)                                                          \
  hipLaunchKernelGGL((beam_topk_kernel<T, KV>), dim3(B), dim3(kBlock), 0,    \
                     stream, reinterpret_cast<const T*>(lp.data_ptr()),      \
                     sc.data_ptr<float>(), out_v.data_ptr<float>(),          \
                     out_i.data_ptr<long>(), B, NB, V, (int)suppress,        \
                     (int)k)
#define RUNT(T)                                                              \
  do {                                                                       \
    if (k <= 4) RUNK(T, 4);                                                  \
    else if (k <= 8) RUNK(T, 8);                                             \
    else RUNK(T, 16);                                                        \
  } while (0)
  if (lp.scalar_type() == torch::kFloat32) RUNT(float);
  else if (lp.scalar_type() == torch::kBFloat16) RUNT(__hip_bfloat16);
  else if (lp.scalar_type() == torch::kHalf) RUNT(__half);
  else TORCH_CHECK(false, "beam_topk: unsupported dtype");
#undef RUNT
#undef RUNK
  return {out_v, out_i};
}
