// Hand-written MFMA weight-gradient GEMM for CDNA4 (gfx950).
//
//   dW [KD, N] = x^T @ dy      x: [M, KD] bf16 rm, dy: [M, N] bf16 rm
//
// This is the GEMM class hipBLASLt loses on for NMT training shapes: the
// reduction runs over M = B*T (65536) while the output is small
// (1024x1024 .. 4096x1024), so without split-K the launch is 16-256
// workgroups on a 256-CU chip. Measured (bench_gemm, r2): torch/TunableOp
// and the full hipBLASLt algo pool both sit at 537-960 TF/s here vs
// 1150-1450 TF/s on the same-FLOP dgrad shapes. This kernel parallelizes
// the M-reduction explicitly: grid.z ranks each reduce a slab of M into
// fp32 partials combined by global atomics (S>1) or written directly
// (S==1).
//
// Structure (guide: cdna_hip_programming.md §5 canonical GEMM + T14):
//   - 128(kd) x 128(n) tile per 256-thread workgroup, BK=64 m-steps;
//   - both operands are staged TRANSPOSED through LDS ([kd][m] / [n][m]
//     images, odd-dword row stride) because the mfma A/B fragments want
//     8 contiguous m-elements per lane while x/dy store m as the row dim;
//     the transpose happens in the register stage (4-row b64 packs);
//   - per wave: 2x2 mfma_f32_32x32x16_bf16 tiles -> 64x64 output quadrant,
//     fp32 accumulators (16 regs x 4);
//   - T14 pipeline: next tile's global loads issue right after the
//     barrier, under the current tile's MFMA work.
// (ref semantics: the dW = x^T dy GEMMs of every Dense backward,
// SURVEY.md §3.3 #23.)
#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

using namespace mamd;

typedef __bf16 bf16x8_w __attribute__((ext_vector_type(8)));
typedef float f32x16_w __attribute__((ext_vector_type(16)));

namespace {

constexpr int BM = 128;   // kd tile
constexpr int BN = 128;   // n tile
constexpr int BK = 64;    // m (reduction) step
// Row stride 72 elems (144 B): every row start is 16B-aligned, so the
// fragment reads are single ds_read_b128 (4 LDS cycles vs 8 for the
// read2_b64 pair the 8B-aligned variant forced). b128 banking is
// (a/4)%64 over 16-lane groups: 36 dwords/row makes rows within a group
// distinct; the m-block XOR swizzle spreads the write phase.
constexpr int LDS_STRIDE = BK + 8;

// Swizzled m-position: 8-element m-blocks XORed by row bits 3..5 so both
// the transposed write phase and the fragment reads are bank-spread while
// 8-element m-runs stay contiguous (guide T2 adapted to a [row][m] image).
__device__ __forceinline__ int swz(int row, int m) {
  return (((m >> 3) ^ ((row >> 3) & 7)) << 3) + (m & 7);
}

__device__ __forceinline__ bf16x8_w lds_read8(const __hip_bfloat16* p) {
  // rows are 16B-aligned (stride 144B) and swizzle offsets are 16B
  // multiples: a true b128 read
  return *reinterpret_cast<const bf16x8_w*>(p);
}

// Pack 4 bf16 (one per source register) into a 64-bit value for the
// transposed LDS write: dst holds m, m+1, m+2, m+3 of one kd/n column.
__device__ __forceinline__ unsigned long long pack4(const bf16x8_w& a,
                                                    const bf16x8_w& b,
                                                    const bf16x8_w& c,
                                                    const bf16x8_w& d,
                                                    int j) {
  union {
    unsigned short u[4];
    unsigned long long v;
  } r;
  union { __bf16 h; unsigned short u; } t;
  t.h = a[j]; r.u[0] = t.u;
  t.h = b[j]; r.u[1] = t.u;
  t.h = c[j]; r.u[2] = t.u;
  t.h = d[j]; r.u[3] = t.u;
  return r.v;
}

template <bool SPLIT>
__global__ __launch_bounds__(256)
void wgrad_kernel(const __hip_bfloat16* __restrict__ X,
                  const __hip_bfloat16* __restrict__ DY,
                  void* __restrict__ OUT,  // bf16 [KD,N] or fp32 [KD,N]
                  float* __restrict__ DB,  // fp32 [N] bias grad, or null
                  long M, long KD, long N, long m_per_split,
                  int tx, int ty, int S) {
  __shared__ __hip_bfloat16 lx[BM][LDS_STRIDE];   // x^T image [kd][m]
  __shared__ __hip_bfloat16 ly[BN][LDS_STRIDE];   // dy^T image [n][m]

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wid = tid >> 6;
  const int hi = lane >> 5;
  const int lq = lane & 31;

  // XCD-pinned kd-tiles (1-D grid; observed: block b runs on XCD b%8,
  // MI355X_MICROARCH §Workgroup dispatch). All ty*S blocks of one
  // kd-tile land on ONE XCD, so the x-slab each m-step needs is read
  // from L3/HBM once and re-served from that XCD's L2 — the 128x128
  // baseline was L3-bound on x re-reads (bench: 465-600 TF/s).
  long kdt, nt, z;
  const long fid = blockIdx.x;
  if ((tx & 7) == 0) {
    const long rest = fid >> 3;
    const long per = (long)ty * S;
    kdt = (fid & 7) + 8 * (rest / per);
    const long r2 = rest % per;
    nt = r2 % ty;
    z = r2 / ty;
  } else {
    kdt = fid % tx;
    nt = (fid / tx) % ty;
    z = fid / ((long)tx * ty);
  }
  const long kd0 = kdt * BM;
  const long n0 = nt * BN;
  const long m_lo = z * m_per_split;
  const long m_hi = min(m_lo + m_per_split, M);

  // wave quadrant: wid -> (kd half, n half), each 64x64
  const int wkd = (wid & 1) * 64;
  const int wn = (wid >> 1) * 64;

  // staging map: thread covers a (4m x 8cols) micro-tile of each operand.
  // col_chunk = tid % 16 (8 columns each), m_grp = tid / 16 (4 rows each)
  const int s_col = (tid & 15) * 8;
  const int s_m = (tid >> 4) * 4;

  f32x16_w acc[2][2];
#pragma unroll
  for (int a = 0; a < 2; ++a)
#pragma unroll
    for (int b = 0; b < 2; ++b) acc[a][b] = (f32x16_w)0.f;

  bf16x8_w xr[4], yr[4];  // staged rows (m, m+1, m+2, m+3)
  // fused bias gradient: kd-tile-0 blocks see every (n, m) exactly once;
  // each thread accumulates its 8 dy columns across the m-slab — the dy
  // values are already in registers for staging, so db costs no traffic
  float dbacc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) dbacc[j] = 0.f;

  // interior tiles take the unconditional path: a per-lane
  // "in-range ? load : zero" select makes hipcc branch around each load and
  // wait vmcnt(0) per element (guide §5.5 trap 4c) — the loads must be
  // straight-line so they pipeline under the MFMA work
  const bool cols_full = (kd0 + BM <= KD) && (n0 + BN <= N);
  auto issue_loads = [&](long m_base) {
    const long kd = kd0 + s_col;
    const long n = n0 + s_col;
    if (cols_full && m_base + BK <= m_hi) {      // wave-uniform branch
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const long m = m_base + s_m + r;
        xr[r] = *reinterpret_cast<const bf16x8_w*>(X + m * KD + kd);
        yr[r] = *reinterpret_cast<const bf16x8_w*>(DY + m * N + n);
      }
      return;
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const long m = m_base + s_m + r;
      const bool mok = m < m_hi;
      xr[r] = (mok && kd + 8 <= KD)
          ? *reinterpret_cast<const bf16x8_w*>(X + m * KD + kd)
          : (bf16x8_w)(__bf16)0.f;
      yr[r] = (mok && n + 8 <= N)
          ? *reinterpret_cast<const bf16x8_w*>(DY + m * N + n)
          : (bf16x8_w)(__bf16)0.f;
      if (mok && kd + 8 > KD) {  // partial kd tail
#pragma unroll
        for (int j = 0; j < 8; ++j)
          xr[r][j] = (kd + j < KD) ? (__bf16)(float)X[m * KD + kd + j]
                                   : (__bf16)0.f;
      }
      if (mok && n + 8 > N) {    // partial n tail
#pragma unroll
        for (int j = 0; j < 8; ++j)
          yr[r][j] = (n + j < N) ? (__bf16)(float)DY[m * N + n + j]
                                 : (__bf16)0.f;
      }
    }
  };

  auto write_tiles = [&]() {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const int rx = s_col + j;
      *reinterpret_cast<unsigned long long*>(&lx[rx][swz(rx, s_m)]) =
          pack4(xr[0], xr[1], xr[2], xr[3], j);
      *reinterpret_cast<unsigned long long*>(&ly[rx][swz(rx, s_m)]) =
          pack4(yr[0], yr[1], yr[2], yr[3], j);
    }
  };

  const bool want_db = (DB != nullptr) && (kdt == 0);
  issue_loads(m_lo);
  write_tiles();
  if (want_db) {
#pragma unroll
    for (int r = 0; r < 4; ++r)
#pragma unroll
      for (int j = 0; j < 8; ++j) dbacc[j] += (float)yr[r][j];
  }
  __syncthreads();

  for (long m0 = m_lo; m0 < m_hi; m0 += BK) {
    const bool last = (m0 + BK >= m_hi);
    if (!last) {
      issue_loads(m0 + BK);  // in flight under this tile's MFMA
    }

#pragma unroll
    for (int ks = 0; ks < BK / 16; ++ks) {
      const int mb = ks * 16 + 8 * hi;
      const int ra0 = wkd + lq, ra1 = wkd + 32 + lq;
      const int rb0 = wn + lq, rb1 = wn + 32 + lq;
      bf16x8_w a0 = lds_read8(&lx[ra0][swz(ra0, mb)]);
      bf16x8_w a1 = lds_read8(&lx[ra1][swz(ra1, mb)]);
      bf16x8_w b0 = lds_read8(&ly[rb0][swz(rb0, mb)]);
      bf16x8_w b1 = lds_read8(&ly[rb1][swz(rb1, mb)]);
      acc[0][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b0, acc[0][0], 0, 0, 0);
      acc[0][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a0, b1, acc[0][1], 0, 0, 0);
      acc[1][0] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b0, acc[1][0], 0, 0, 0);
      acc[1][1] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a1, b1, acc[1][1], 0, 0, 0);
    }
    if (!last) {
      __syncthreads();       // MFMA done reading the current images
      write_tiles();         // stage the prefetched tile
      if (want_db) {
#pragma unroll
        for (int r = 0; r < 4; ++r)
#pragma unroll
          for (int j = 0; j < 8; ++j) dbacc[j] += (float)yr[r][j];
      }
      __syncthreads();
    }
  }
  if (want_db) {
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      const long n = n0 + s_col + j;
      if (n < N) atomicAdd(&DB[n], dbacc[j]);
    }
  }

  // Epilogue. C-fragment: row = kd_local = crow(r, hi) (+ wkd + 32*a),
  // col = n_local = lq (+ wn + 32*b). mfma(A=x^T frag, B=dy frag) gives
  // D[i=kd][j=n] with the standard map: lane j = lq, rows over regs.
#pragma unroll
  for (int a = 0; a < 2; ++a) {
#pragma unroll
    for (int b = 0; b < 2; ++b) {
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const long kd = kd0 + wkd + 32 * a + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const long n = n0 + wn + 32 * b + lq;
        if (kd < KD && n < N) {
          if constexpr (SPLIT) {
            atomicAdd(reinterpret_cast<float*>(OUT) + kd * N + n,
                      acc[a][b][r]);
          } else {
            reinterpret_cast<__hip_bfloat16*>(OUT)[kd * N + n] =
                (__hip_bfloat16)acc[a][b][r];
          }
        }
      }
    }
  }
}

}  // namespace

// dW [KD, N] bf16 = x^T @ dy (+ fp32 db = colsum(dy) fused, free since
// dy is already in registers for staging). `splits` 0 = auto.
std::vector<torch::Tensor> wgrad_bias_bf16(torch::Tensor x, torch::Tensor dy,
                                           int64_t splits, bool want_db) {
  TORCH_CHECK(x.is_cuda() && x.dim() == 2 && dy.dim() == 2 &&
              x.scalar_type() == torch::kBFloat16 &&
              dy.scalar_type() == torch::kBFloat16 &&
              x.size(0) == dy.size(0));
  auto xc = x.contiguous();
  auto dyc = dy.contiguous();
  const long M = xc.size(0), KD = xc.size(1), N = dyc.size(1);
  const long tx = (KD + BM - 1) / BM, ty = (N + BN - 1) / BN;
  long S = splits;
  if (S <= 0) {
    // enough blocks to fill 256 CUs a couple of times over, without
    // blowing up the fp32 atomic traffic on large outputs
    S = std::max(1L, std::min(16L, 512L / std::max(1L, tx * ty)));
    while (S > 1 && (M + S - 1) / S < BK) S /= 2;
  }
  const long m_per = ((M + S - 1) / S + BK - 1) / BK * BK;
  auto stream = at::cuda::getCurrentHIPStream();
  dim3 grid((unsigned)(tx * ty * S));
  torch::Tensor db;
  float* dbp = nullptr;
  if (want_db) {
    db = torch::zeros({N}, xc.options().dtype(torch::kFloat32));
    dbp = db.data_ptr<float>();
  } else {
    db = torch::empty({0}, xc.options());
  }
  if (S == 1) {
    auto out = torch::empty({KD, N}, xc.options());
    hipLaunchKernelGGL((wgrad_kernel<false>), grid, dim3(256), 0, stream,
                       reinterpret_cast<const __hip_bfloat16*>(xc.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(dyc.data_ptr()),
                       out.data_ptr(), dbp, M, KD, N, m_per,
                       (int)tx, (int)ty, (int)S);
    return {out, db};
  }
  auto ws = torch::zeros({KD, N}, xc.options().dtype(torch::kFloat32));
  hipLaunchKernelGGL((wgrad_kernel<true>), grid, dim3(256), 0, stream,
                     reinterpret_cast<const __hip_bfloat16*>(xc.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(dyc.data_ptr()),
                     ws.data_ptr(), dbp, M, KD, N, m_per,
                     (int)tx, (int)ty, (int)S);
  return {ws.to(torch::kBFloat16), db};
}

torch::Tensor wgrad_bf16(torch::Tensor x, torch::Tensor dy, int64_t splits) {
  return wgrad_bias_bf16(x, dy, splits, false)[0];
}
