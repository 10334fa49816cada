// MFMA bf16 GEMM for the transformer linear layers (MI355X / gfx950).
//
// Reference ops this replaces: QKV/out projections,
// GEGLU FeedForward matmuls, proj_in/out of the spatial transformer —
// the rocBLAS/Tensile share of the SD-2.1 finetune step
// (/root/reference/diff_train.py:644 runs them through diffusers linears).
//
// One kernel template covers all three passes of torch.nn.Linear without
// materializing any transpose, because each operand can be staged from
// either a k-contiguous ("row-major") or a k-strided ("col-major") layout:
//
//   C[M,N] = sum_k A(m,k) * B(n,k)   (+ bias[n])        [out rows from A,
//                                                         out cols from B]
//   forward : A = x  [M,K]  (TA=0)   B = W  [N,K]  (TB=0)   + bias
//   dgrad   : A = dy [M,N'] (TA=0)   B = W' = mem[N'][K] read k-strided
//             (TB=1: contraction n', out-col k)         -> dx
//   wgrad   : A = dy^T (TA=1: mem [M][N], contraction m) and
//             B = x^T  (TB=1: mem [M][K])               -> dW (+ dbias,
//             the column-sum of dy fused into the A staging pass)
//
// Structure: the validated conv v3 discipline — 128x128 tile, 4 waves as
// 2x2 of 64x64 sub-tiles, v_mfma_f32_16x16x32_bf16, double-buffered LDS
// with register staging (next tile's global loads issued before the MFMA
// loop, write pass after it, ONE barrier per K-step), LDS pitch BK+8 so
// 16-lane ds_read_b128 groups land on 16 distinct banks. Split-K over the
// contraction (fp32 atomics + finalize) when the M/N grid starves 256 CUs.

#include "dcr_common.h"

namespace dcr_gemm {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

using bf16_t = __hip_bfloat16;

// Stage one operand tile [128 rows][BK k] into registers.
//  - !T : memory is row-major [rows][K] (k contiguous): 2 threads/row,
//         each loads BK/2 contiguous bf16 as uint4s.
//  - T  : memory is [K][rows] (rows contiguous): each thread owns one row
//         r = tid&127 and k-half (tid>>7); BK/2 coalesced 2-byte loads
//         (consecutive lanes -> consecutive rows), packed to uint4s.
// Out-of-range rows/k are zero-filled (contraction tail, ragged M/N).
template <int BK, bool T, bool COLSUM>
struct Stager {
  static constexpr int NCH = BK / 16;  // uint4s per thread (BK/2 bf16)
  const bf16_t* __restrict__ src;
  long ld;        // row stride (!T: = K) or r stride (T: = rows_total)
  long row0;      // first row of the tile
  long rows;      // total rows of the operand (M or N)
  int K;          // contraction extent
  int tid;
  float colsum;   // running sum of this thread's loaded values (wgrad dbias)

  __device__ void init(const bf16_t* s, long ld_, long row0_, long rows_,
                       int K_, int tid_) {
    src = s; ld = ld_; row0 = row0_; rows = rows_; K = K_; tid = tid_;
    colsum = 0.f;
  }
  __device__ int st_row() const { return T ? (tid & 127) : (tid >> 1); }
  __device__ int st_k() const {
    return T ? ((tid >> 7) * (BK / 2)) : ((tid & 1) * (BK / 2));
  }

  __device__ void load(int k0, uint4 (&v)[NCH]) {
#pragma unroll
    for (int t = 0; t < NCH; ++t) v[t] = make_uint4(0, 0, 0, 0);
    const long r = row0 + st_row();
    const int kb = k0 + st_k();
    if (r >= rows) return;
    if (!T) {
      const bf16_t* p = src + r * ld + kb;
      if (kb + BK / 2 <= K) {
#pragma unroll
        for (int t = 0; t < NCH; ++t)
          v[t] = reinterpret_cast<const uint4*>(p)[t];
      } else {
#pragma unroll
        for (int t = 0; t < NCH; ++t)
          if (kb + t * 8 + 8 <= K) v[t] = reinterpret_cast<const uint4*>(p)[t];
          else if (kb + t * 8 < K) {  // ragged 8-tail: scalar fill
            ushort tmp[8] = {0, 0, 0, 0, 0, 0, 0, 0};
            for (int e = 0; e < 8 && kb + t * 8 + e < K; ++e)
              tmp[e] = reinterpret_cast<const ushort*>(p)[t * 8 + e];
            v[t] = *reinterpret_cast<const uint4*>(tmp);
          }
      }
    } else {
      // k-strided gather: BK/2 coalesced scalar loads down the k column
      ushort tmp[BK / 2];
      const bf16_t* p = src + (long)kb * ld + r;
#pragma unroll
      for (int j = 0; j < BK / 2; ++j) {
        tmp[j] = (kb + j < K) ? reinterpret_cast<const ushort*>(p)[(long)j * ld]
                              : (ushort)0;
      }
      if (COLSUM) {
#pragma unroll
        for (int j = 0; j < BK / 2; ++j) {
          __hip_bfloat16 h = *reinterpret_cast<__hip_bfloat16*>(&tmp[j]);
          colsum += __bfloat162float(h);
        }
      }
#pragma unroll
      for (int t = 0; t < NCH; ++t)
        v[t] = *reinterpret_cast<const uint4*>(&tmp[t * 8]);
    }
  }

  template <int PITCH>
  __device__ void store(short* lds, const uint4 (&v)[NCH]) {
    uint4* d = reinterpret_cast<uint4*>(lds + st_row() * PITCH + st_k());
#pragma unroll
    for (int t = 0; t < NCH; ++t) d[t] = v[t];
  }
};

template <int BK, bool TA, bool TB, bool DBIAS>
__global__ __launch_bounds__(256)
void gemm_bf16_kernel(const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
                      const float* __restrict__ bias, bf16_t* __restrict__ C,
                      float* __restrict__ ws, float* __restrict__ dbias,
                      long M, long N, int K, int splitz) {
  constexpr int PITCH = BK + 8;
  __shared__ short sA[2][128 * PITCH];
  __shared__ short sB[2][128 * PITCH];

  const long m0 = (long)blockIdx.x * 128;
  const long n0 = (long)blockIdx.y * 128;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int kgrp = lane >> 4;
  const int wr = (wid >> 1) * 64;
  const int wc = (wid & 1) * 64;

  f32x4_t acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  Stager<BK, TA, DBIAS> stA;
  Stager<BK, TB, false> stB;
  stA.init(A, TA ? M : (long)K, m0, M, K, (int)threadIdx.x);
  stB.init(B, TB ? N : (long)K, n0, N, K, (int)threadIdx.x);

  const int nsteps = (K + BK - 1) / BK;
  const int spz = (nsteps + splitz - 1) / splitz;
  const int step0 = blockIdx.z * spz;
  const int step1 = min(nsteps, step0 + spz);
  constexpr int NCH = BK / 16;

  uint4 av[NCH], bv[NCH];
  if (step0 < step1) {
    stA.load(step0 * BK, av);
    stB.load(step0 * BK, bv);
    stA.template store<PITCH>(sA[0], av);
    stB.template store<PITCH>(sB[0], bv);
  }
  __syncthreads();

  int cur = 0;
  for (int step = step0; step < step1; ++step) {
    const bool more = (step + 1 < step1);
    if (more) {
      stA.load((step + 1) * BK, av);
      stB.load((step + 1) * BK, bv);
    }

#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        af[i] = *reinterpret_cast<const bf16x8*>(
            sA[cur] + (wr + i * 16 + l16) * PITCH + kk * 32 + kgrp * 8);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bf[j] = *reinterpret_cast<const bf16x8*>(
            sB[cur] + (wc + j * 16 + l16) * PITCH + kk * 32 + kgrp * 8);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bf[j],
                                                              acc[i][j], 0, 0, 0);
    }

    if (more) {
      stA.template store<PITCH>(sA[cur ^ 1], av);
      stB.template store<PITCH>(sB[cur ^ 1], bv);
      __syncthreads();
      cur ^= 1;
    }
  }

  // fused dbias (wgrad): A is dy^T, each staging thread has summed its
  // loaded dy values over the contraction; one tile of blocks (y==0)
  // covers every (m, n) pair exactly once per z-slice.
  if (DBIAS && blockIdx.y == 0) {
    const long r = m0 + stA.st_row();
    if (r < M && stA.colsum != 0.f) atomicAdd(&dbias[r], stA.colsum);
  }

#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const long m = m0 + wr + i * 16 + kgrp * 4 + rr;
      if (m >= M) continue;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const long n = n0 + wc + j * 16 + l16;
        if (n >= N) continue;
        if (splitz > 1) {
          atomicAdd(&ws[m * N + n], acc[i][j][rr]);
        } else {
          float v = acc[i][j][rr] + (bias ? bias[n] : 0.f);
          C[m * N + n] = __float2bfloat16(v);
        }
      }
    }
  }
}

__global__ void gemm_finalize_kernel(const float* __restrict__ ws,
                                     const float* __restrict__ bias,
                                     bf16_t* __restrict__ y, long total,
                                     long N) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    float v = ws[i] + (bias ? bias[i % N] : 0.f);
    y[i] = __float2bfloat16(v);
  }
}

}  // namespace dcr_gemm

#include "dcr_launchers.h"

namespace dcr {

void gemm_bf16_launch(const void* A, const void* B, const float* bias,
                      void* C, float* ws, float* dbias, long M, long N, int K,
                      int ta, int tb, int want_dbias, int splitz,
                      hipStream_t st) {
  dim3 grid((unsigned)((M + 127) / 128), (unsigned)((N + 127) / 128),
            (unsigned)splitz),
      block(256);
  const auto* a = (const dcr_gemm::bf16_t*)A;
  const auto* b = (const dcr_gemm::bf16_t*)B;
  auto* c = (dcr_gemm::bf16_t*)C;

#define DCR_GEMM_LAUNCH(BK, TA, TB, DB)                                        \
  hipLaunchKernelGGL((dcr_gemm::gemm_bf16_kernel<BK, TA, TB, DB>), grid,       \
                     block, 0, st, a, b, bias, c, ws, dbias, M, N, K, splitz)

  const bool bk64 = (K % 64 == 0) || (K > 256);
  if (!ta && !tb) {
    if (bk64) DCR_GEMM_LAUNCH(64, false, false, false);
    else      DCR_GEMM_LAUNCH(32, false, false, false);
  } else if (!ta && tb) {
    if (bk64) DCR_GEMM_LAUNCH(64, false, true, false);
    else      DCR_GEMM_LAUNCH(32, false, true, false);
  } else if (ta && tb && want_dbias) {
    if (bk64) DCR_GEMM_LAUNCH(64, true, true, true);
    else      DCR_GEMM_LAUNCH(32, true, true, true);
  } else {
    if (bk64) DCR_GEMM_LAUNCH(64, true, true, false);
    else      DCR_GEMM_LAUNCH(32, true, true, false);
  }
#undef DCR_GEMM_LAUNCH

  if (splitz > 1) {
    long total = M * N;
    long blk = (total / 4 + 255) / 256;
    if (blk > 8192) blk = 8192;
    hipLaunchKernelGGL(dcr_gemm::gemm_finalize_kernel, dim3((unsigned)blk),
                       dim3(256), 0, st, ws, bias, c, total, N);
  }
}

}  // namespace dcr
