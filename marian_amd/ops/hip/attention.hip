// Fused flash-style attention forward for CDNA4 (gfx950), bf16, Dh = 64.
//
// ref semantics: the reference composes attention from ProdBatched + mask +
// gSoftmax + ProdBatched (src/models/transformer.h LayerAttention +
// src/tensors/gpu/prod.cpp). Rebuilt here as ONE kernel per the MI355X
// guide's transposed-fragment recipe:
//
//  - S^T = mfma_f32_32x32x16_bf16(A=K-tile, B=Q^T)  -> lane holds one
//    q-COLUMN of S^T, so the softmax row-reduce (over keys) is 16 in-lane
//    values + one __shfl_xor(32) with the partner half: no LDS round-trip.
//  - online softmax (running m, l per q) entirely in registers; the O
//    accumulator is kept TRANSPOSED (O^T fragments), so the per-q rescale
//    factor is a per-lane scalar multiply.
//  - P (C-layout) is repacked to the MFMA B-operand layout with
//    v_cvt_pk_bf16_f32 + __builtin_amdgcn_permlane32_swap (guide T12).
//  - V is staged through LDS TRANSPOSED (V^T image) so the PV A-fragment is
//    one 16-byte ds_read per 8 keys.
//  - masks are PREFIX masks (NMT pads at the end): kv_len per batch entry +
//    optional causal, evaluated in-register - no mask tensor traffic.
//
// One wave (64 lanes) per 32-row q-block; grid = (ceil(Tq/32), B*H).
// Saves per-row logsumexp for a later fused backward.
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

using namespace mamd;

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x16 __attribute__((ext_vector_type(16)));

namespace {

constexpr float NEG_BIG = -1e30f;

// C-fragment row for reg r, half hi (mfma_f32_32x32x16_bf16):
// row = (r&3) + 8*(r>>2) + 4*hi
__device__ __forceinline__ int crow(int r, int hi) {
  return (r & 3) + 8 * (r >> 2) + 4 * hi;
}

__device__ __forceinline__ unsigned cvt_pk_bf16(float lo, float hi) {
  unsigned r;
  asm volatile("v_cvt_pk_bf16_f32 %0, %1, %2" : "=v"(r) : "v"(lo), "v"(hi));
  return r;
}

// Repack a 32x32 S^T C-fragment (16 f32, already exp()ed) into the two
// B-operand bf16x8 fragments of the PV mfma (k-chunks 0-15 and 16-31).
// See guide T12: r = permlane32_swap(cvtpk(p[4i],p[4i+1]), cvtpk(p[4i+2],p[4i+3]))
__device__ __forceinline__ void pack_p_to_b(const float p[16], bf16x8 out[2]) {
#pragma unroll
  for (int c = 0; c < 2; ++c) {   // k-chunk: regs 8c..8c+7 = k rows 16c..16c+15
    unsigned w[4];
    // pairs of consecutive k rows within this chunk:
    // regs (8c+0,8c+1) -> rows (16c+0,1)+4hi ; (8c+2,8c+3) -> (16c+2,3)+4hi
    // regs (8c+4,8c+5) -> rows (16c+8,9)+4hi ; (8c+6,8c+7) -> (16c+10,11)+4hi
    unsigned a01 = cvt_pk_bf16(p[8 * c + 0], p[8 * c + 1]);  // k 0,1 | 4,5
    unsigned a23 = cvt_pk_bf16(p[8 * c + 2], p[8 * c + 3]);  // k 2,3 | 6,7
    unsigned a89 = cvt_pk_bf16(p[8 * c + 4], p[8 * c + 5]);  // k 8,9 | 12,13
    unsigned aAB = cvt_pk_bf16(p[8 * c + 6], p[8 * c + 7]);  // k 10,11 | 14,15
    {
      auto r = __builtin_amdgcn_permlane32_swap(a01, a89, false, false);
      w[0] = r[0];  // lane<32: k0,1 ; lane>=32: k8,9
      w[2] = r[1];  // lane<32: k4,5 ; lane>=32: k12,13
    }
    {
      auto r = __builtin_amdgcn_permlane32_swap(a23, aAB, false, false);
      w[1] = r[0];  // lane<32: k2,3 ; lane>=32: k10,11
      w[3] = r[1];  // lane<32: k6,7 ; lane>=32: k14,15
    }
    // B fragment k-order: dwords k(0,1),(2,3),(4,5),(6,7) (+8 on upper half)
    union {
      unsigned u[4];
      bf16x8 v;
    } cvt;
    cvt.u[0] = w[0];
    cvt.u[1] = w[1];
    cvt.u[2] = w[2];
    cvt.u[3] = w[3];
    out[c] = cvt.v;
  }
}

// Element strides for a [B, H, T, D] tensor view (innermost dim must be
// contiguous). Lets [B,T,H*D] activations feed the kernels through permuted
// VIEWS - no transpose copies around attention.
struct TStride {
  long b, h, s;  // batch, head, sequence strides (elements)
};

// LDS V^T image: [D][KVBLK] with row pad to dodge bank conflicts.
constexpr int KVBLK = 32;
constexpr int QBLK = 32;
constexpr int VT_PAD = 2;  // elements; row stride = odd dword count
// (N+2 bf16 elems = odd number of 4B banks: stride 34 elem = 17 dwords,
// 66 elem = 33 dwords -> consecutive rows walk all 64 LDS banks; the old
// pad 8 gave 20/36-dword strides, gcd 4 with 64 banks = 4-way conflicts)

// Forward kernel: NW waves per workgroup, each owning one 32-row q-tile of
// the same (b,h); the KVBLK=64 K/V^T LDS stage is shared by all waves.
template <int D, int NW, bool CAUSAL, bool HAS_LEN>
__global__ __launch_bounds__(64 * NW)
void attn_fwd_kernel(const __hip_bfloat16* __restrict__ Q,
                     const __hip_bfloat16* __restrict__ K,
                     const __hip_bfloat16* __restrict__ V,
                     const int* __restrict__ kv_lens,  // [B] or null
                     __hip_bfloat16* __restrict__ O,
                     float* __restrict__ LSE,          // [B*H*Tq] or null
                     TStride qs, TStride ks, TStride os,
                     int B, int H, int Tq, int Tk, float scale) {
  constexpr int KVB = 64;                       // kv tile (2 x 32-row mfma)
  __shared__ __hip_bfloat16 vt[D][KVB + VT_PAD];  // V^T image (shared)
  __shared__ float obounce[NW][QBLK][D + 5];  // odd dword row stride (bank-conflict free)

  const int lane = threadIdx.x & 63;
  const int tid = threadIdx.x;
  const int wid = tid >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  const int bh = blockIdx.y;
  const int b = bh / H;
  const int h = bh % H;
  const int q0 = (blockIdx.x * NW + wid) * QBLK;
  const bool wave_live = q0 < Tq;
  const int kv_len = HAS_LEN ? kv_lens[b] : Tk;

  const __hip_bfloat16* Qb = Q + (size_t)b * qs.b + (size_t)h * qs.h;
  const __hip_bfloat16* Kb = K + (size_t)b * ks.b + (size_t)h * ks.h;
  const __hip_bfloat16* Vb = V + (size_t)b * ks.b + (size_t)h * ks.h;
  __hip_bfloat16* Ob = O + (size_t)b * os.b + (size_t)h * os.h;

  // Q^T B-fragments, scaled once
  bf16x8 qfrag[D / 16];
  {
    const int qrow = wave_live ? min(q0 + lq, Tq - 1) : 0;
    const __hip_bfloat16* qp = Qb + (size_t)qrow * qs.s;
#pragma unroll
    for (int c = 0; c < D / 16; ++c) {
      qfrag[c] = *reinterpret_cast<const bf16x8*>(qp + 16 * c + 8 * hi);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        qfrag[c][j] = (__bf16)((float)qfrag[c][j] * scale);
    }
  }

  float m_run = NEG_BIG;
  float l_run = 0.f;
  f32x16 ot[D / 32];
#pragma unroll
  for (int dt = 0; dt < D / 32; ++dt) ot[dt] = (f32x16)0.f;

  // shared-staging loop bound: the LAST wave's causal frontier
  const int q_hi = min((blockIdx.x * NW + NW) * QBLK, Tq);
  const int kv_end_blk = CAUSAL ? min(kv_len, q_hi) : kv_len;
  const int kv_end_own = CAUSAL ? min(kv_len, q0 + QBLK) : kv_len;
  const int q_abs = q0 + lq;

  // async-STAGE split (guide T14): each thread keeps its share of the NEXT
  // V tile in registers; global loads are issued right after the barrier so
  // HBM latency hides under the tile's MFMA compute, and the ds_write of
  // the prefetched tile happens at the top of the next iteration.
  constexpr int RPP = (NW * 64) / (D / 8);  // staged rows per pass
  constexpr int NPASS = (KVB + RPP - 1) / RPP;
  bf16x8 vld[NPASS];
  const int stage_row = tid / (D / 8);      // k-row this thread stages
  const int stage_col = (tid % (D / 8)) * 8;

  auto issue_v_loads = [&](int k0) {
#pragma unroll
    for (int ps = 0; ps < NPASS; ++ps) {
      const int kl = ps * RPP + stage_row;
      const int krow = k0 + kl;
      vld[ps] = (kl < KVB && krow < kv_len)
          ? *reinterpret_cast<const bf16x8*>(Vb + (size_t)krow * ks.s + stage_col)
          : (bf16x8)(__bf16)0.f;
    }
  };
  issue_v_loads(0);

  for (int k0 = 0; k0 < kv_end_blk; k0 += KVB) {
    // ---- write the prefetched V^T tile (transpose in flight) -----------
    {
#pragma unroll
      for (int ps = 0; ps < NPASS; ++ps) {
        const int kl = ps * RPP + stage_row;
        if (kl < KVB) {
#pragma unroll
          for (int j = 0; j < 8; ++j)
            vt[stage_col + j][kl] = (__hip_bfloat16)(float)vld[ps][j];
        }
      }
    }
    __builtin_amdgcn_s_waitcnt(0xc07f);  // lgkmcnt(0) only: keep loads live
    __builtin_amdgcn_s_barrier();
    if (k0 + KVB < kv_end_blk)
      issue_v_loads(k0 + KVB);  // in flight under this tile's compute

    if (wave_live && k0 < kv_end_own) {
      // ---- S^T = K.Q^T for both 32-row k-halves -------------------------
      f32x16 st0 = (f32x16)0.f, st1 = (f32x16)0.f;
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        const int kr0 = min(k0 + lq, Tk - 1);
        const int kr1 = min(k0 + 32 + lq, Tk - 1);
        bf16x8 kf0 = *reinterpret_cast<const bf16x8*>(
            Kb + (size_t)kr0 * ks.s + 16 * c + 8 * hi);
        bf16x8 kf1 = *reinterpret_cast<const bf16x8*>(
            Kb + (size_t)kr1 * ks.s + 16 * c + 8 * hi);
        st0 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf0, qfrag[c], st0, 0, 0, 0);
        st1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf1, qfrag[c], st1, 0, 0, 0);
      }

      // ---- mask + tile max ----------------------------------------------
      float p[32];
      float tile_max = NEG_BIG;
#pragma unroll
      for (int half = 0; half < 2; ++half) {
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          const int k_abs = k0 + 32 * half + crow(r, hi);
          float v = half ? st1[r] : st0[r];
          const bool dead = (k_abs >= kv_len) || (CAUSAL && k_abs > q_abs);
          v = dead ? NEG_BIG : v;
          p[16 * half + r] = v;
          tile_max = fmaxf(tile_max, v);
        }
      }
      tile_max = fmaxf(tile_max, __shfl_xor(tile_max, 32));

      // ---- online rescale -----------------------------------------------
      const float m_new = fmaxf(m_run, tile_max);
      const float alpha = (m_new > NEG_BIG * 0.5f) ? __expf(m_run - m_new) : 1.f;
      float psum = 0.f;
#pragma unroll
      for (int r = 0; r < 32; ++r) {
        p[r] = (p[r] <= NEG_BIG * 0.5f) ? 0.f : __expf(p[r] - m_new);
        psum += p[r];
      }
      psum += __shfl_xor(psum, 32);
      l_run = l_run * alpha + psum;
      m_run = m_new;
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
        for (int r = 0; r < 16; ++r) ot[dt][r] *= alpha;

      // ---- O^T += V^T . P^T over 4 k-chunks -----------------------------
      bf16x8 pb0[2], pb1[2];
      pack_p_to_b(p, pb0);
      pack_p_to_b(p + 16, pb1);
#pragma unroll
      for (int dt = 0; dt < D / 32; ++dt) {
#pragma unroll
        for (int c = 0; c < 2; ++c) {
          bf16x8 va0 = *reinterpret_cast<const bf16x8*>(
              &vt[32 * dt + lq][16 * c + 8 * hi]);
          bf16x8 va1 = *reinterpret_cast<const bf16x8*>(
              &vt[32 * dt + lq][32 + 16 * c + 8 * hi]);
          ot[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va0, pb0[c],
                                                           ot[dt], 0, 0, 0);
          ot[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(va1, pb1[c],
                                                           ot[dt], 0, 0, 0);
        }
      }
    }
    __builtin_amdgcn_s_barrier();
  }

  // ---- epilogue ----------------------------------------------------------
  if (wave_live) {
    const float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
#pragma unroll
    for (int dt = 0; dt < D / 32; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r)
        obounce[wid][lq][32 * dt + crow(r, hi)] = ot[dt][r] * inv_l;
  }
  __builtin_amdgcn_s_barrier();
  if (wave_live) {
    const int lanes_per_row = D / 8;
    const int rows_per_pass = 64 / lanes_per_row;
    for (int qq = 0; qq < QBLK; qq += rows_per_pass) {
      const int qrow = q0 + qq + lane / lanes_per_row;
      const int dcol = (lane % lanes_per_row) * 8;
      if (qrow < Tq) {
        bf16x8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          out[j] = (__bf16)obounce[wid][qq + lane / lanes_per_row][dcol + j];
        *reinterpret_cast<bf16x8*>(Ob + (size_t)qrow * os.s + dcol) = out;
      }
    }
    if (LSE != nullptr && hi == 0) {
      const int qrow = q0 + lq;
      if (qrow < Tq)
        LSE[(size_t)bh * Tq + qrow] = m_run + __logf(fmaxf(l_run, 1e-30f));
    }
  }
}

}  // namespace

// Strided-view support: a [B,H,T,D] view whose last dim is contiguous and
// 16B-aligned passes through un-copied (e.g. a permute of [B,T,H*D]).
static bool view_ok(const torch::Tensor& t) {
  return t.stride(3) == 1 && (t.stride(2) % 8 == 0) && (t.stride(1) % 8 == 0)
      && (t.stride(0) % 8 == 0);
}

static TStride tstride(const torch::Tensor& t) {
  return TStride{t.stride(0), t.stride(1), t.stride(2)};
}

// q,k,v: [B, H, T, D] bf16 (any strided view, D contiguous).
// returns (o [B,H,T,D] view with BSHD storage, lse)
std::vector<torch::Tensor> attention_fwd(torch::Tensor q, torch::Tensor k,
                                         torch::Tensor v,
                                         c10::optional<torch::Tensor> kv_lens,
                                         bool causal, double scale,
                                         bool need_lse) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  TORCH_CHECK(q.dim() == 4 && k.dim() == 4 && v.dim() == 4);
  auto qc = view_ok(q) ? q : q.contiguous();
  auto kc = view_ok(k) ? k : k.contiguous();
  auto vc = (v.strides() == kc.strides() && view_ok(v)) ? v : v.contiguous();
  if (vc.strides() != kc.strides()) {  // kernels share k/v strides
    kc = kc.contiguous();
    vc = vc.contiguous();
  }
  const int B = qc.size(0), H = qc.size(1), Tq = qc.size(2), D = qc.size(3);
  const int Tk = kc.size(2);
  TORCH_CHECK(kc.size(0) == B && kc.size(1) == H && kc.size(3) == D);
  TORCH_CHECK(D == 64, "attention_fwd: only Dh=64 supported (got ", D, ")");
  // O allocated in [B, T, H, D] storage, returned as a [B,H,T,D] view so
  // the caller's head-join is a free reshape
  auto o = torch::empty({B, Tq, H, (int)D}, qc.options()).permute({0, 2, 1, 3});
  torch::Tensor lse;
  float* lse_ptr = nullptr;
  if (need_lse) {
    lse = torch::empty({B, H, Tq}, qc.options().dtype(torch::kFloat32));
    lse_ptr = lse.data_ptr<float>();
  } else {
    lse = torch::Tensor();
  }
  const int* lens_ptr = nullptr;
  torch::Tensor lensc;
  if (kv_lens.has_value() && kv_lens->defined()) {
    lensc = kv_lens->to(torch::kInt32).contiguous();
    TORCH_CHECK(lensc.numel() == B);
    lens_ptr = lensc.data_ptr<int>();
  }
  auto stream = at::cuda::getCurrentHIPStream();
  const int nqt = (Tq + 31) / 32;
  const int NW = (nqt % 4 == 0) ? 4 : (nqt % 2 == 0 ? 2 : 1);
  dim3 grid(nqt / NW, B * H);
#define RUN1(NWV, CAUSAL, HASLEN)                                            \
  hipLaunchKernelGGL((attn_fwd_kernel<64, NWV, CAUSAL, HASLEN>), grid,       \
                     dim3(64 * NWV), 0, stream,                              \
                     reinterpret_cast<const __hip_bfloat16*>(qc.data_ptr()), \
                     reinterpret_cast<const __hip_bfloat16*>(kc.data_ptr()), \
                     reinterpret_cast<const __hip_bfloat16*>(vc.data_ptr()), \
                     lens_ptr,                                               \
                     reinterpret_cast<__hip_bfloat16*>(o.data_ptr()),        \
                     lse_ptr, tstride(qc), tstride(kc), tstride(o),          \
                     B, H, Tq, Tk, (float)scale)
#define RUN(CAUSAL, HASLEN)                                                  \
  do {                                                                       \
    if (NW == 4) RUN1(4, CAUSAL, HASLEN);                                    \
    else if (NW == 2) RUN1(2, CAUSAL, HASLEN);                               \
    else RUN1(1, CAUSAL, HASLEN);                                            \
  } while (0)
  if (causal && lens_ptr) RUN(true, true);
  else if (causal) RUN(true, false);
  else if (lens_ptr) RUN(false, true);
  else RUN(false, false);
#undef RUN
#undef RUN1
  return {o, lse};
}

// ===========================================================================
// Fused attention BACKWARD (flash-attn-2 style split, bf16, Dh=64).
//
// Two kernels, no atomics:
//  - attn_bwd_dq_kernel:  parallel over q-tiles, loops kv; dQ^T accumulates
//    in registers. Uses the "q per lane" orientation (S^T = mfma(K, Q^T),
//    dP^T = mfma(V, dO^T)); dS^T packed via cvt_pk+permlane32_swap feeds
//    dQ^T += mfma(K^T, dS^T) with K^T staged transposed in LDS.
//  - attn_bwd_dkv_kernel: parallel over kv-tiles, loops q; dK/dV accumulate
//    in registers. Uses the "k per lane" orientation (S = mfma(Q, K^T),
//    dP = mfma(dO, V^T)); packed P^T / dS^T feed
//    dV += mfma(P^T, dO), dK += mfma(dS^T, Q) with Q^T/dO^T staged in LDS.
// D = rowsum(dO o O) is precomputed by the host wrapper.
// ===========================================================================

namespace {

template <int D, int NW, bool CAUSAL, bool HAS_LEN>
__global__ __launch_bounds__(64 * NW)
void attn_bwd_dq_kernel(const __hip_bfloat16* __restrict__ Q,
                        const __hip_bfloat16* __restrict__ K,
                        const __hip_bfloat16* __restrict__ V,
                        const __hip_bfloat16* __restrict__ dO,
                        const float* __restrict__ LSE,
                        const float* __restrict__ Dsum,  // [B*H*Tq]
                        const int* __restrict__ kv_lens,
                        __hip_bfloat16* __restrict__ dQ,
                        TStride qs, TStride ks, TStride dos, TStride dqs,
                        int B, int H, int Tq, int Tk, float scale) {
  // K^T image staged 2 tiles at a time: one cooperative pass + one
  // barrier pair per 64 kv rows instead of per 32 (T<=64 shapes are
  // staging/barrier-bound: 313us/call at B1024 T64 vs ~0.5us of MFMA)
  __shared__ __hip_bfloat16 kt[D][2 * KVBLK + VT_PAD];
  __shared__ float dq_b[NW][QBLK][D + 5];  // odd dword row stride (bank-conflict free)

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  const int bh = blockIdx.y;
  const int b = bh / H;
  const int q0 = (blockIdx.x * NW + wid) * QBLK;
  const bool wave_live = q0 < Tq;
  const int kv_len = HAS_LEN ? kv_lens[b] : Tk;
  const int h = bh % H;
  const __hip_bfloat16* Qb = Q + (size_t)b * qs.b + (size_t)h * qs.h;
  const __hip_bfloat16* Kb = K + (size_t)b * ks.b + (size_t)h * ks.h;
  const __hip_bfloat16* Vb = V + (size_t)b * ks.b + (size_t)h * ks.h;
  const __hip_bfloat16* dOb = dO + (size_t)b * dos.b + (size_t)h * dos.h;
  __hip_bfloat16* dQb = dQ + (size_t)b * dqs.b + (size_t)h * dqs.h;

  bf16x8 qfrag[D / 16], dofrag[D / 16];
  {
    const bool qlive = (q0 + lq) < Tq;
    const int qrow = qlive ? (q0 + lq) : (Tq - 1);
    const __hip_bfloat16* qp = Qb + (size_t)qrow * qs.s;
    const __hip_bfloat16* dp = dOb + (size_t)qrow * dos.s;
#pragma unroll
    for (int c = 0; c < D / 16; ++c) {
      qfrag[c] = *reinterpret_cast<const bf16x8*>(qp + 16 * c + 8 * hi);
      bf16x8 dv = *reinterpret_cast<const bf16x8*>(dp + 16 * c + 8 * hi);
      if (!qlive) dv = (bf16x8)(__bf16)0.f;
      dofrag[c] = dv;
    }
  }
  const int q_abs = q0 + lq;
  const float lse_lane = (wave_live && q_abs < Tq)
      ? LSE[(size_t)bh * Tq + q_abs] : 0.f;
  const float dsum_lane = (wave_live && q_abs < Tq)
      ? Dsum[(size_t)bh * Tq + q_abs] : 0.f;

  f32x16 dqt[D / 32];
#pragma unroll
  for (int dt2 = 0; dt2 < D / 32; ++dt2) dqt[dt2] = (f32x16)0.f;

  const int q_hi = min((blockIdx.x * NW + NW) * QBLK, Tq);
  const int kv_end_blk = CAUSAL ? min(kv_len, q_hi) : kv_len;
  const int kv_end = CAUSAL ? min(kv_len, q0 + QBLK) : kv_len;
  constexpr int STW = 2 * KVBLK;   // staging width (2 inner tiles)
  for (int k0s = 0; k0s < kv_end_blk; k0s += STW) {
    {
      // cooperative K^T staging with all NW*64 threads
      const int rows_per_pass = (NW * 64) / (D / 8);
      for (int kk = 0; kk < STW; kk += rows_per_pass) {
        const int kl = kk + tid / (D / 8);
        const int krow2 = k0s + kl;
        const int dcol = (tid % (D / 8)) * 8;
        if (kl < STW) {
          bf16x8 kv = (krow2 < kv_len)
              ? *reinterpret_cast<const bf16x8*>(Kb + (size_t)krow2 * ks.s + dcol)
              : (bf16x8)(__bf16)0.f;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            kt[dcol + j][kl] = (__hip_bfloat16)(float)kv[j];
        }
      }
    }
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_s_barrier();
    const int k_tile_hi = min(k0s + STW, kv_end_blk);
    for (int k0 = k0s; k0 < k_tile_hi; k0 += KVBLK) {
    if (!wave_live || k0 >= kv_end)
      continue;
    bf16x8 kfrag[D / 16], vfrag[D / 16];
    {
      const int krow = min(k0 + lq, Tk - 1);
      const __hip_bfloat16* kp = Kb + (size_t)krow * ks.s;
      const __hip_bfloat16* vp = Vb + (size_t)krow * ks.s;
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        kfrag[c] = *reinterpret_cast<const bf16x8*>(kp + 16 * c + 8 * hi);
        vfrag[c] = *reinterpret_cast<const bf16x8*>(vp + 16 * c + 8 * hi);
      }
    }

    f32x16 st = (f32x16)0.f;
    f32x16 dpt = (f32x16)0.f;
#pragma unroll
    for (int c = 0; c < D / 16; ++c) {
      st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfrag[c], qfrag[c], st, 0, 0, 0);
      dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfrag[c], dofrag[c], dpt, 0, 0, 0);
    }
    float ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int k_abs = k0 + crow(r, hi);
      const bool dead = (k_abs >= kv_len) || (CAUSAL && k_abs > q_abs)
                        || (q_abs >= Tq);
      const float pval = dead ? 0.f : __expf(st[r] * scale - lse_lane);
      ds[r] = pval * (dpt[r] - dsum_lane) * scale;
    }
    bf16x8 dst_op[2];
    pack_p_to_b(ds, dst_op);
#pragma unroll
    for (int dt2 = 0; dt2 < D / 32; ++dt2) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        bf16x8 ka = *reinterpret_cast<const bf16x8*>(
            &kt[32 * dt2 + lq][(k0 - k0s) + 16 * c + 8 * hi]);
        dqt[dt2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ka, dst_op[c],
                                                           dqt[dt2], 0, 0, 0);
      }
    }
    }  // inner k0 tiles
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: dQ via per-wave LDS bounce slice
  if (wave_live) {
#pragma unroll
    for (int dt2 = 0; dt2 < D / 32; ++dt2)
#pragma unroll
      for (int r = 0; r < 16; ++r)
        dq_b[wid][lq][32 * dt2 + crow(r, hi)] = dqt[dt2][r];
  }
  __builtin_amdgcn_s_barrier();
  if (wave_live) {
    const int lanes_per_row = D / 8;
    const int rows_per_pass = 64 / lanes_per_row;
    for (int qq = 0; qq < QBLK; qq += rows_per_pass) {
      const int qrow = q0 + qq + lane / lanes_per_row;
      const int dcol = (lane % lanes_per_row) * 8;
      if (qrow < Tq) {
        bf16x8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          out[j] = (__bf16)dq_b[wid][qq + lane / lanes_per_row][dcol + j];
        *reinterpret_cast<bf16x8*>(dQb + (size_t)qrow * dqs.s + dcol) = out;
      }
    }
  }
}

template <int D, int NW, bool CAUSAL, bool HAS_LEN>
__global__ __launch_bounds__(64 * NW)
void attn_bwd_dkv_kernel(const __hip_bfloat16* __restrict__ Q,
                         const __hip_bfloat16* __restrict__ K,
                         const __hip_bfloat16* __restrict__ V,
                         const __hip_bfloat16* __restrict__ dO,
                         const float* __restrict__ LSE,
                         const float* __restrict__ Dsum,
                         const int* __restrict__ kv_lens,
                         __hip_bfloat16* __restrict__ dK,
                         __hip_bfloat16* __restrict__ dV,
                         TStride qs, TStride ks, TStride dos, TStride dks,
                         int B, int H, int Tq, int Tk, float scale) {
  // Q^T / dO^T images staged 2 tiles at a time (see dq kernel note)
  __shared__ __hip_bfloat16 qt[D][2 * QBLK + VT_PAD];
  __shared__ __hip_bfloat16 dot[D][2 * QBLK + VT_PAD];
  __shared__ float kv_b[NW][KVBLK][D + 5];  // odd dword row stride (bank-conflict free)

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;
  const int bh = blockIdx.y;
  const int b = bh / H;
  const int k0 = (blockIdx.x * NW + wid) * KVBLK;
  const bool wave_live = k0 < Tk;
  const int kv_len = HAS_LEN ? kv_lens[b] : Tk;
  const int h = bh % H;
  const __hip_bfloat16* Qb = Q + (size_t)b * qs.b + (size_t)h * qs.h;
  const __hip_bfloat16* Kb = K + (size_t)b * ks.b + (size_t)h * ks.h;
  const __hip_bfloat16* Vb = V + (size_t)b * ks.b + (size_t)h * ks.h;
  const __hip_bfloat16* dOb = dO + (size_t)b * dos.b + (size_t)h * dos.h;
  __hip_bfloat16* dKb = dK + (size_t)b * dks.b + (size_t)h * dks.h;
  __hip_bfloat16* dVb = dV + (size_t)b * dks.b + (size_t)h * dks.h;

  bf16x8 kfrag[D / 16], vfrag[D / 16];
  {
    const int krow = min(k0 + lq, Tk - 1);
    const __hip_bfloat16* kp = Kb + (size_t)krow * ks.s;
    const __hip_bfloat16* vp = Vb + (size_t)krow * ks.s;
#pragma unroll
    for (int c = 0; c < D / 16; ++c) {
      kfrag[c] = *reinterpret_cast<const bf16x8*>(kp + 16 * c + 8 * hi);
      vfrag[c] = *reinterpret_cast<const bf16x8*>(vp + 16 * c + 8 * hi);
    }
  }
  const int k_lane = k0 + lq;

  f32x16 dk_acc[D / 32], dv_acc[D / 32];
#pragma unroll
  for (int dt2 = 0; dt2 < D / 32; ++dt2) {
    dk_acc[dt2] = (f32x16)0.f;
    dv_acc[dt2] = (f32x16)0.f;
  }

  // causal: the FIRST wave's kv tile bounds the shared q loop start
  const int k0_first = (blockIdx.x * NW) * KVBLK;
  const int q_start_blk = CAUSAL ? (k0_first / QBLK) * QBLK : 0;
  const int q_start_own = CAUSAL ? (k0 / QBLK) * QBLK : 0;
  constexpr int STW = 2 * QBLK;    // staging width (2 inner tiles)
  const int q_start_stage = CAUSAL ? (k0_first / STW) * STW : 0;
  for (int q0s = q_start_stage; q0s < Tq; q0s += STW) {
    {
      // cooperative Q^T/dO^T staging with all NW*64 threads
      const int rows_per_pass = (NW * 64) / (D / 8);
      for (int qq = 0; qq < STW; qq += rows_per_pass) {
        const int qlocal = qq + tid / (D / 8);
        const int qrow2 = q0s + qlocal;
        const int dcol = (tid % (D / 8)) * 8;
        if (qlocal < STW) {
          const bool live = qrow2 < Tq;
          const int qr = live ? qrow2 : (Tq - 1);
          bf16x8 vq = *reinterpret_cast<const bf16x8*>(Qb + (size_t)qr * qs.s + dcol);
          bf16x8 vd = *reinterpret_cast<const bf16x8*>(dOb + (size_t)qr * dos.s + dcol);
          if (!live) {
            vq = (bf16x8)(__bf16)0.f;
            vd = (bf16x8)(__bf16)0.f;
          }
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            qt[dcol + j][qlocal] = (__hip_bfloat16)(float)vq[j];
            dot[dcol + j][qlocal] = (__hip_bfloat16)(float)vd[j];
          }
        }
      }
    }
    __builtin_amdgcn_s_waitcnt(0);
    __builtin_amdgcn_s_barrier();
    const int q_tile_hi = min(q0s + STW, Tq);
    for (int q0 = max(q0s, q_start_blk); q0 < q_tile_hi; q0 += QBLK) {
    if (!wave_live || q0 < q_start_own)
      continue;
    // per-q-tile fragments (direct loads)
    bf16x8 qfrag[D / 16], dofrag[D / 16];
    {
      const bool qlive = (q0 + lq) < Tq;
      const int qrow = qlive ? (q0 + lq) : (Tq - 1);
      const __hip_bfloat16* qp = Qb + (size_t)qrow * qs.s;
      const __hip_bfloat16* dp = dOb + (size_t)qrow * dos.s;
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        qfrag[c] = *reinterpret_cast<const bf16x8*>(qp + 16 * c + 8 * hi);
        bf16x8 dv = *reinterpret_cast<const bf16x8*>(dp + 16 * c + 8 * hi);
        if (!qlive) dv = (bf16x8)(__bf16)0.f;
        dofrag[c] = dv;
      }
    }

    f32x16 s = (f32x16)0.f;
    f32x16 dp = (f32x16)0.f;
#pragma unroll
    for (int c = 0; c < D / 16; ++c) {
      s = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qfrag[c], kfrag[c], s, 0, 0, 0);
      dp = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dofrag[c], vfrag[c], dp, 0, 0, 0);
    }
    float pv[16], ds[16];
#pragma unroll
    for (int r = 0; r < 16; ++r) {
      const int q_r = q0 + crow(r, hi);
      const bool qlive = q_r < Tq;
      const float lse_r = qlive ? LSE[(size_t)bh * Tq + q_r] : 0.f;
      const float dsum_r = qlive ? Dsum[(size_t)bh * Tq + q_r] : 0.f;
      const bool dead = (k_lane >= kv_len) || (CAUSAL && k_lane > q_r) || !qlive;
      const float pval = dead ? 0.f : __expf(s[r] * scale - lse_r);
      pv[r] = pval;
      ds[r] = pval * (dp[r] - dsum_r) * scale;
    }
    bf16x8 p_op[2], ds_op[2];
    pack_p_to_b(pv, p_op);
    pack_p_to_b(ds, ds_op);
#pragma unroll
    for (int dt2 = 0; dt2 < D / 32; ++dt2) {
#pragma unroll
      for (int c = 0; c < 2; ++c) {
        bf16x8 dob = *reinterpret_cast<const bf16x8*>(
            &dot[32 * dt2 + lq][(q0 - q0s) + 16 * c + 8 * hi]);
        bf16x8 qb = *reinterpret_cast<const bf16x8*>(
            &qt[32 * dt2 + lq][(q0 - q0s) + 16 * c + 8 * hi]);
        dv_acc[dt2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(p_op[c], dob,
                                                              dv_acc[dt2], 0, 0, 0);
        dk_acc[dt2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ds_op[c], qb,
                                                              dk_acc[dt2], 0, 0, 0);
      }
    }
    }  // inner q0 tiles
    __builtin_amdgcn_s_barrier();
  }

  // epilogue: dV then dK via per-wave LDS bounce slices
  for (int which = 0; which < 2; ++which) {
    __hip_bfloat16* out_base = (which == 0) ? dVb : dKb;
    f32x16* acc = (which == 0) ? dv_acc : dk_acc;
    if (wave_live) {
#pragma unroll
      for (int dt2 = 0; dt2 < D / 32; ++dt2)
#pragma unroll
        for (int r = 0; r < 16; ++r)
          kv_b[wid][crow(r, hi)][32 * dt2 + lq] = acc[dt2][r];
    }
    __builtin_amdgcn_s_barrier();
    if (wave_live) {
      const int lanes_per_row = D / 8;
      const int rows_per_pass = 64 / lanes_per_row;
      for (int kk = 0; kk < KVBLK; kk += rows_per_pass) {
        const int krow = k0 + kk + lane / lanes_per_row;
        const int dcol = (lane % lanes_per_row) * 8;
        if (krow < Tk) {
          bf16x8 out;
          const bool live = krow < kv_len;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            out[j] = live
                ? (__bf16)kv_b[wid][kk + lane / lanes_per_row][dcol + j]
                : (__bf16)0.f;
          *reinterpret_cast<bf16x8*>(out_base + (size_t)krow * dks.s + dcol) = out;
        }
      }
    }
    __builtin_amdgcn_s_barrier();
  }
}

// Merged small-T backward: for Tq,Tk <= 64 the split dQ/dKV kernels are
// launch/staging-bound (91 TF vs 201 TF at T512). One 4-wave block per
// (b,h): waves 0/1 own the two kv-tiles (dK/dV), waves 2/3 the two
// q-tiles (dQ); Q^T, dO^T and K^T images are staged once and shared, the
// dO*O rowsum is computed into LDS by the q-waves (no dsum kernel, no
// global dsum tensor), and the fp32 epilogue bounce aliases the image
// memory — two barriers total.
template <int D, bool CAUSAL, bool HAS_LEN>
__global__ __launch_bounds__(256)
void attn_bwd_fused64_kernel(const __hip_bfloat16* __restrict__ Q,
                             const __hip_bfloat16* __restrict__ K,
                             const __hip_bfloat16* __restrict__ V,
                             const __hip_bfloat16* __restrict__ O,
                             const __hip_bfloat16* __restrict__ dO,
                             const float* __restrict__ LSE,
                             const int* __restrict__ kv_lens,
                             __hip_bfloat16* __restrict__ dQ,
                             __hip_bfloat16* __restrict__ dK,
                             __hip_bfloat16* __restrict__ dV,
                             TStride qs, TStride ks, TStride dos,
                             TStride oss, TStride dqs, TStride dks,
                             int B, int H, int Tq, int Tk, float scale) {
  constexpr int TW = 64;                  // staged width
  constexpr int IMG = D * (TW + VT_PAD);  // one bf16 image
  constexpr int BNC = 32 * (D + 5);       // one wave's fp32 bounce slice
  __shared__ __align__(16) char smem[4 * BNC * 4 > 3 * IMG * 2
                                     ? 4 * BNC * 4 : 3 * IMG * 2];
  __shared__ float dsum_sh[TW];
  __shared__ float lse_sh[TW];
  auto qt = [&](int r, int c) -> __hip_bfloat16& {
    return reinterpret_cast<__hip_bfloat16*>(smem)[r * (TW + VT_PAD) + c];
  };
  auto dot_ = [&](int r, int c) -> __hip_bfloat16& {
    return reinterpret_cast<__hip_bfloat16*>(smem)[IMG + r * (TW + VT_PAD) + c];
  };
  auto kt = [&](int r, int c) -> __hip_bfloat16& {
    return reinterpret_cast<__hip_bfloat16*>(smem)[2 * IMG + r * (TW + VT_PAD) + c];
  };
  auto bounce = [&](int w, int r, int c) -> float& {
    return reinterpret_cast<float*>(smem)[w * BNC + r * (D + 5) + c];
  };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;               // 0,1: kv role; 2,3: q role
  const int hi = lane >> 5;
  const int lq = lane & 31;
  const int bh = blockIdx.x;
  const int b = bh / H;
  const int h = bh % H;
  const int kv_len = HAS_LEN ? kv_lens[b] : Tk;
  const __hip_bfloat16* Qb = Q + (size_t)b * qs.b + (size_t)h * qs.h;
  const __hip_bfloat16* Kb = K + (size_t)b * ks.b + (size_t)h * ks.h;
  const __hip_bfloat16* Vb = V + (size_t)b * ks.b + (size_t)h * ks.h;
  const __hip_bfloat16* Ob = O + (size_t)b * oss.b + (size_t)h * oss.h;
  const __hip_bfloat16* dOb = dO + (size_t)b * dos.b + (size_t)h * dos.h;

  // ---- stage Q^T, dO^T, K^T cooperatively (256 threads, 8 lanes/row) --
  {
    const int rows_per_pass = 256 / (D / 8);          // 32
    const int rl = tid / (D / 8);
    const int dcol = (tid % (D / 8)) * 8;
    for (int rr = 0; rr < TW; rr += rows_per_pass) {
      const int r = rr + rl;
      const bool ql = r < Tq;
      const bool kl = r < kv_len && r < Tk;
      bf16x8 vq = ql ? *reinterpret_cast<const bf16x8*>(
                           Qb + (size_t)r * qs.s + dcol)
                     : (bf16x8)(__bf16)0.f;
      bf16x8 vd = ql ? *reinterpret_cast<const bf16x8*>(
                           dOb + (size_t)r * dos.s + dcol)
                     : (bf16x8)(__bf16)0.f;
      bf16x8 vk = kl ? *reinterpret_cast<const bf16x8*>(
                           Kb + (size_t)r * ks.s + dcol)
                     : (bf16x8)(__bf16)0.f;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        qt(dcol + j, r) = (__hip_bfloat16)(float)vq[j];
        dot_(dcol + j, r) = (__hip_bfloat16)(float)vd[j];
        kt(dcol + j, r) = (__hip_bfloat16)(float)vk[j];
      }
    }
  }

  // role tile base: kv waves own kv rows, q waves own q rows
  const int t0 = (wid & 1) * 32;
  bf16x8 rfrag[D / 16], dofrag[D / 16];   // role-row fragments
  const bool q_role = wid >= 2;
  {
    const int row = t0 + lq;
    const int lim = q_role ? Tq : min(kv_len, Tk);
    const int rr = min(row, (q_role ? Tq : Tk) - 1);
    const __hip_bfloat16* p0 = q_role ? (Qb + (size_t)rr * qs.s)
                                      : (Kb + (size_t)rr * ks.s);
    const __hip_bfloat16* p1 = q_role ? (dOb + (size_t)rr * dos.s)
                                      : (Vb + (size_t)rr * ks.s);
    const bool live = row < lim;
#pragma unroll
    for (int c = 0; c < D / 16; ++c) {
      bf16x8 v0 = *reinterpret_cast<const bf16x8*>(p0 + 16 * c + 8 * hi);
      bf16x8 v1 = *reinterpret_cast<const bf16x8*>(p1 + 16 * c + 8 * hi);
      if (!live) {
        v0 = (bf16x8)(__bf16)0.f;
        v1 = (bf16x8)(__bf16)0.f;
      }
      rfrag[c] = v0;     // q-role: Q row; kv-role: K row
      dofrag[c] = v1;    // q-role: dO row; kv-role: V row
    }
    if (q_role) {
      // dsum + lse for my q row into LDS (shared with the kv waves)
      float ds = 0.f;
      const __hip_bfloat16* op = Ob + (size_t)rr * oss.s;
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        const bf16x8 ov = *reinterpret_cast<const bf16x8*>(op + 16 * c + 8 * hi);
#pragma unroll
        for (int j = 0; j < 8; ++j) ds += (float)dofrag[c][j] * (float)ov[j];
      }
      ds += __shfl_xor(ds, 32, 64);
      if (live && hi == 0) {
        dsum_sh[row] = ds;
        lse_sh[row] = LSE[(size_t)bh * Tq + row];
      }
    }
  }
  __builtin_amdgcn_s_waitcnt(0);
  __builtin_amdgcn_s_barrier();

  // ---- compute ------------------------------------------------------
  f32x16 acc0[D / 32], acc1[D / 32];      // q: dQ,- ; kv: dV,dK
#pragma unroll
  for (int d2 = 0; d2 < D / 32; ++d2) {
    acc0[d2] = (f32x16)0.f;
    acc1[d2] = (f32x16)0.f;
  }

  if (q_role) {
    const int q0 = t0;
    const int q_abs = q0 + lq;
    const float lse_l = (q_abs < Tq) ? lse_sh[q_abs] : 0.f;
    const float ds_l = (q_abs < Tq) ? dsum_sh[q_abs] : 0.f;
    const int kv_end = CAUSAL ? min(kv_len, q0 + 32) : kv_len;
    for (int k0 = 0; k0 < kv_end; k0 += 32) {
      bf16x8 kfrag[D / 16], vfrag[D / 16];
      const int krow = min(k0 + lq, Tk - 1);
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        kfrag[c] = *reinterpret_cast<const bf16x8*>(
            Kb + (size_t)krow * ks.s + 16 * c + 8 * hi);
        vfrag[c] = *reinterpret_cast<const bf16x8*>(
            Vb + (size_t)krow * ks.s + 16 * c + 8 * hi);
      }
      f32x16 st = (f32x16)0.f, dpt = (f32x16)0.f;
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kfrag[c], rfrag[c], st, 0, 0, 0);
        dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vfrag[c], dofrag[c], dpt, 0, 0, 0);
      }
      float ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int k_abs = k0 + crow(r, hi);
        const bool dead = (k_abs >= kv_len) || (CAUSAL && k_abs > q_abs)
                          || (q_abs >= Tq);
        const float pval = dead ? 0.f : __expf(st[r] * scale - lse_l);
        ds[r] = pval * (dpt[r] - ds_l) * scale;
      }
      bf16x8 dst_op[2];
      pack_p_to_b(ds, dst_op);
#pragma unroll
      for (int d2 = 0; d2 < D / 32; ++d2)
#pragma unroll
        for (int c = 0; c < 2; ++c) {
          bf16x8 ka = *reinterpret_cast<const bf16x8*>(
              &kt(32 * d2 + lq, k0 + 16 * c + 8 * hi));
          acc0[d2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              ka, dst_op[c], acc0[d2], 0, 0, 0);
        }
    }
  } else {
    const int k0 = t0;
    const int k_lane = k0 + lq;
    const int q_start = CAUSAL ? (k0 / 32) * 32 : 0;
    for (int q0 = q_start; q0 < Tq; q0 += 32) {
      bf16x8 qfr[D / 16], dofr[D / 16];
      const int qrow = min(q0 + lq, Tq - 1);
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        qfr[c] = *reinterpret_cast<const bf16x8*>(
            Qb + (size_t)qrow * qs.s + 16 * c + 8 * hi);
        bf16x8 dv = *reinterpret_cast<const bf16x8*>(
            dOb + (size_t)qrow * dos.s + 16 * c + 8 * hi);
        if (q0 + lq >= Tq) dv = (bf16x8)(__bf16)0.f;
        dofr[c] = dv;
      }
      f32x16 st = (f32x16)0.f, dpt = (f32x16)0.f;
#pragma unroll
      for (int c = 0; c < D / 16; ++c) {
        st = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qfr[c], rfrag[c], st, 0, 0, 0);
        dpt = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dofr[c], dofrag[c], dpt, 0, 0, 0);
      }
      float pv[16], ds[16];
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int q_r = q0 + crow(r, hi);
        const bool qlive = q_r < Tq;
        const float lse_r = qlive ? lse_sh[q_r] : 0.f;
        const float dsum_r = qlive ? dsum_sh[q_r] : 0.f;
        const bool dead = (k_lane >= kv_len) || (CAUSAL && k_lane > q_r)
                          || !qlive;
        const float pval = dead ? 0.f : __expf(st[r] * scale - lse_r);
        pv[r] = pval;
        ds[r] = pval * (dpt[r] - dsum_r) * scale;
      }
      bf16x8 p_op[2], ds_op[2];
      pack_p_to_b(pv, p_op);
      pack_p_to_b(ds, ds_op);
#pragma unroll
      for (int d2 = 0; d2 < D / 32; ++d2)
#pragma unroll
        for (int c = 0; c < 2; ++c) {
          bf16x8 dob = *reinterpret_cast<const bf16x8*>(
              &dot_(32 * d2 + lq, q0 + 16 * c + 8 * hi));
          bf16x8 qb = *reinterpret_cast<const bf16x8*>(
              &qt(32 * d2 + lq, q0 + 16 * c + 8 * hi));
          acc0[d2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              p_op[c], dob, acc0[d2], 0, 0, 0);
          acc1[d2] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
              ds_op[c], qb, acc1[d2], 0, 0, 0);
        }
    }
  }
  __builtin_amdgcn_s_barrier();   // images dead; bounce aliases them

  // ---- epilogue: per-wave fp32 bounce -> bf16 rows -------------------
  __hip_bfloat16* dKb = dK + (size_t)b * dks.b + (size_t)h * dks.h;
  __hip_bfloat16* dVb = dV + (size_t)b * dks.b + (size_t)h * dks.h;
  __hip_bfloat16* dQb = dQ + (size_t)b * dqs.b + (size_t)h * dqs.h;
  const int npass = q_role ? 1 : 2;       // kv waves: dV then dK
  for (int which = 0; which < npass; ++which) {
    f32x16* acc = (which == 0) ? acc0 : acc1;
    if (q_role) {
      // dQ accumulator layout: lane lq = q row, crow = d rows
#pragma unroll
      for (int d2 = 0; d2 < D / 32; ++d2)
#pragma unroll
        for (int r = 0; r < 16; ++r)
          bounce(wid, lq, 32 * d2 + crow(r, hi)) = acc[d2][r];
    } else {
#pragma unroll
      for (int d2 = 0; d2 < D / 32; ++d2)
#pragma unroll
        for (int r = 0; r < 16; ++r)
          bounce(wid, crow(r, hi), 32 * d2 + lq) = acc[d2][r];
    }
    const int lanes_per_row = D / 8;
    const int rows_per_pass = 64 / lanes_per_row;
    for (int rr = 0; rr < 32; rr += rows_per_pass) {
      const int local = rr + lane / lanes_per_row;
      const int row = t0 + local;
      const int dcol = (lane % lanes_per_row) * 8;
      __hip_bfloat16* out_base =
          q_role ? dQb : ((which == 0) ? dVb : dKb);
      const TStride& os_ = q_role ? dqs : dks;
      const int lim = q_role ? Tq : Tk;
      const bool live = row < lim && (q_role || row < kv_len);
      if (row < lim) {
        bf16x8 out;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          out[j] = live ? (__bf16)bounce(wid, local, dcol + j) : (__bf16)0.f;
        *reinterpret_cast<bf16x8*>(out_base + (size_t)row * os_.s + dcol) = out;
      }
    }
  }
}

// Fused delta = rowsum(dO * O) in one pass (bf16 reads, fp32 out).
// The torch composite (two fp32 upcasts + mul + a dim=-1 reduce) cost
// ~680us per call at B1024; this is a single ~45us streaming kernel.
template <int D>
__global__ void attn_dsum_kernel(const __hip_bfloat16* __restrict__ dO,
                                 const __hip_bfloat16* __restrict__ O,
                                 float* __restrict__ dsum,
                                 TStride ds, TStride os,
                                 int H, int Tq, long nrows) {
  constexpr int LPR = D / 8;  // lanes per row, 8B (8 x bf16) each
  const long row = ((long)blockIdx.x * blockDim.x + threadIdx.x) / LPR;
  if (row >= nrows) return;
  const int part = threadIdx.x % LPR;
  const int t = (int)(row % Tq);
  const int h = (int)((row / Tq) % H);
  const long b = row / ((long)Tq * H);
  const bf16x8 d8 = *reinterpret_cast<const bf16x8*>(
      dO + b * ds.b + h * ds.h + (long)t * ds.s + part * 8);
  const bf16x8 o8 = *reinterpret_cast<const bf16x8*>(
      O + b * os.b + h * os.h + (long)t * os.s + part * 8);
  float acc = 0.f;
#pragma unroll
  for (int j = 0; j < 8; ++j) acc += (float)d8[j] * (float)o8[j];
#pragma unroll
  for (int w = 1; w < LPR; w <<= 1) acc += __shfl_xor(acc, w, 64);
  if (part == 0) dsum[row] = acc;
}

}  // namespace

// returns (dq, dk, dv) - all bf16, allocated in [B,T,H,D] storage and
// returned as [B,H,T,D] views (caller reshapes for free)
std::vector<torch::Tensor> attention_bwd_into(
    torch::Tensor q, torch::Tensor k, torch::Tensor v, torch::Tensor o,
    torch::Tensor d_o, torch::Tensor lse, c10::optional<torch::Tensor> kv_lens,
    bool causal, double scale,
    torch::Tensor dq, torch::Tensor dk, torch::Tensor dv) {
  TORCH_CHECK(q.is_cuda() && q.scalar_type() == torch::kBFloat16);
  auto qc = view_ok(q) ? q : q.contiguous();
  auto kc = view_ok(k) ? k : k.contiguous();
  auto vc = (v.strides() == kc.strides() && view_ok(v)) ? v : v.contiguous();
  if (vc.strides() != kc.strides()) {
    kc = kc.contiguous();
    vc = vc.contiguous();
  }
  auto oc = view_ok(o) ? o : o.contiguous();
  auto doc = view_ok(d_o) ? d_o : d_o.contiguous();
  const int B = qc.size(0), H = qc.size(1), Tq = qc.size(2), D = qc.size(3);
  const int Tk = kc.size(2);
  TORCH_CHECK(D == 64, "attention_bwd: only Dh=64 supported");
  TORCH_CHECK(view_ok(dq) && view_ok(dk) && view_ok(dv) &&
              dk.strides() == dv.strides(),
              "attention_bwd: grad outputs must be 8-aligned strided views");
  const int* lens_ptr = nullptr;
  torch::Tensor lensc;
  if (kv_lens.has_value() && kv_lens->defined()) {
    lensc = kv_lens->to(torch::kInt32).contiguous();
    lens_ptr = lensc.data_ptr<int>();
  }
  const int* lens_ptr_early = lens_ptr;
  auto stream_f = at::cuda::getCurrentHIPStream();
  // Merged-kernel path: measured NEUTRAL non-causal (237us vs 235) and
  // SLOWER causal (385 vs 229) at B512 T64 — the dual-role accumulators
  // push past the register budget. Kept opt-in for round-2 iteration.
  static const bool use_fused64 = getenv("MAMD_FUSED_BWD64") != nullptr;
  if (use_fused64 && Tq <= 64 && Tk <= 64) {
    // merged one-launch backward for the NMT training shapes
    dim3 gf(B * H);
    const bool has_len = lens_ptr_early != nullptr;
#define RUNF(CAUSAL, HASLEN)                                                  \
    hipLaunchKernelGGL((attn_bwd_fused64_kernel<64, CAUSAL, HASLEN>), gf,     \
                       dim3(256), 0, stream_f,                                \
                       reinterpret_cast<const __hip_bfloat16*>(qc.data_ptr()),\
                       reinterpret_cast<const __hip_bfloat16*>(kc.data_ptr()),\
                       reinterpret_cast<const __hip_bfloat16*>(vc.data_ptr()),\
                       reinterpret_cast<const __hip_bfloat16*>(oc.data_ptr()),\
                       reinterpret_cast<const __hip_bfloat16*>(doc.data_ptr()),\
                       lse.data_ptr<float>(), lens_ptr_early,                 \
                       reinterpret_cast<__hip_bfloat16*>(dq.data_ptr()),      \
                       reinterpret_cast<__hip_bfloat16*>(dk.data_ptr()),      \
                       reinterpret_cast<__hip_bfloat16*>(dv.data_ptr()),      \
                       tstride(qc), tstride(kc), tstride(doc), tstride(oc),   \
                       tstride(dq), tstride(dk), B, H, Tq, Tk, (float)scale)
    if (causal) {
      if (has_len) RUNF(true, true);
      else RUNF(true, false);
    } else {
      if (has_len) RUNF(false, true);
      else RUNF(false, false);
    }
#undef RUNF
    return {dq, dk, dv};
  }
  // D = rowsum(dO o O) per query row, fp32 — single fused pass
  auto dsum = torch::empty({B, H, Tq},
                           qc.options().dtype(torch::kFloat32));
  {
    const long nrows = (long)B * H * Tq;
    const int blk = 256;
    const long nthreads = nrows * (64 / 8);
    auto stream0 = at::cuda::getCurrentHIPStream();
    hipLaunchKernelGGL((attn_dsum_kernel<64>),
                       dim3((nthreads + blk - 1) / blk), dim3(blk), 0,
                       stream0,
                       reinterpret_cast<const __hip_bfloat16*>(doc.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(oc.data_ptr()),
                       dsum.data_ptr<float>(), tstride(doc), tstride(oc),
                       H, Tq, nrows);
  }
  auto stream = at::cuda::getCurrentHIPStream();
  const int nqt = (Tq + 31) / 32;
  const int nkt = (Tk + 31) / 32;
  int NWQ = (nqt % 4 == 0) ? 4 : (nqt % 2 == 0 ? 2 : 1);
  int NWK = (nkt % 4 == 0) ? 4 : (nkt % 2 == 0 ? 2 : 1);
  if (const char* e = getenv("MAMD_BWD_NW")) {   // A/B experiments
    NWQ = NWK = atoi(e);
    if (nqt % NWQ) NWQ = 1;
    if (nkt % NWK) NWK = 1;
  }
  dim3 gq(nqt / NWQ, B * H);
  dim3 gk(nkt / NWK, B * H);
#define RUNB1(NQ, NK, CAUSAL, HASLEN)                                         \
  do {                                                                        \
    hipLaunchKernelGGL((attn_bwd_dq_kernel<64, NQ, CAUSAL, HASLEN>), gq,      \
                       dim3(64 * NQ), 0, stream,                              \
                       reinterpret_cast<const __hip_bfloat16*>(qc.data_ptr()),\
                       reinterpret_cast<const __hip_bfloat16*>(kc.data_ptr()),\
                       reinterpret_cast<const __hip_bfloat16*>(vc.data_ptr()),\
                       reinterpret_cast<const __hip_bfloat16*>(doc.data_ptr()),\
                       lse.data_ptr<float>(), dsum.data_ptr<float>(),         \
                       lens_ptr,                                              \
                       reinterpret_cast<__hip_bfloat16*>(dq.data_ptr()),      \
                       tstride(qc), tstride(kc), tstride(doc), tstride(dq),   \
                       B, H, Tq, Tk, (float)scale);                           \
    hipLaunchKernelGGL((attn_bwd_dkv_kernel<64, NK, CAUSAL, HASLEN>), gk,     \
                       dim3(64 * NK), 0, stream,                              \
                       reinterpret_cast<const __hip_bfloat16*>(qc.data_ptr()),\
                       reinterpret_cast<const __hip_bfloat16*>(kc.data_ptr()),\
                       reinterpret_cast<const __hip_bfloat16*>(vc.data_ptr()),\
                       reinterpret_cast<const __hip_bfloat16*>(doc.data_ptr()),\
                       lse.data_ptr<float>(), dsum.data_ptr<float>(),         \
                       lens_ptr,                                              \
                       reinterpret_cast<__hip_bfloat16*>(dk.data_ptr()),      \
                       reinterpret_cast<__hip_bfloat16*>(dv.data_ptr()),      \
                       tstride(qc), tstride(kc), tstride(doc), tstride(dk),   \
                       B, H, Tq, Tk, (float)scale);                           \
  } while (0)
#define RUNB2(NQ, CAUSAL, HASLEN)                                             \
  do {                                                                        \
    if (NWK == 4) RUNB1(NQ, 4, CAUSAL, HASLEN);                               \
    else if (NWK == 2) RUNB1(NQ, 2, CAUSAL, HASLEN);                          \
    else RUNB1(NQ, 1, CAUSAL, HASLEN);                                        \
  } while (0)
#define RUNB(CAUSAL, HASLEN)                                                  \
  do {                                                                        \
    if (NWQ == 4) RUNB2(4, CAUSAL, HASLEN);                                   \
    else if (NWQ == 2) RUNB2(2, CAUSAL, HASLEN);                              \
    else RUNB2(1, CAUSAL, HASLEN);                                            \
  } while (0)
  if (causal && lens_ptr) RUNB(true, true);
  else if (causal) RUNB(true, false);
  else if (lens_ptr) RUNB(false, true);
  else RUNB(false, false);
#undef RUNB
#undef RUNB2
#undef RUNB1
  return {dq, dk, dv};
}

// returns (dq, dk, dv) allocated in [B,T,H,D] storage, [B,H,T,D] views
std::vector<torch::Tensor> attention_bwd(torch::Tensor q, torch::Tensor k,
                                         torch::Tensor v, torch::Tensor o,
                                         torch::Tensor d_o, torch::Tensor lse,
                                         c10::optional<torch::Tensor> kv_lens,
                                         bool causal, double scale) {
  const int B = q.size(0), H = q.size(1), Tq = q.size(2), D = q.size(3);
  const int Tk = k.size(2);
  auto dq = torch::empty({B, Tq, H, D}, q.options()).permute({0, 2, 1, 3});
  auto dk = torch::empty({B, Tk, H, D}, q.options()).permute({0, 2, 1, 3});
  auto dv = torch::empty({B, Tk, H, D}, q.options()).permute({0, 2, 1, 3});
  return attention_bwd_into(q, k, v, o, d_o, lse, kv_lens, causal, scale,
                            dq, dk, dv);
}
