// Implicit-GEMM convolution BACKWARD kernels (NHWC, bf16, MFMA) — drafts
// for round 2: compile-verified, not dispatched (ops/conv.py uses
// aten::convolution_backward until these are validated & measured;
// env-gated tests: DCR_NATIVE_CONV_BWD=1).
//
// bwd-weight: dW[k][r][s][c] = sum_{n,p,q} dy[n,p,q,k] * x[n, p*st+r-pad,
//   q*st+s-pad, c].  GEMM: M = K, N = R*S*C, contraction over NPQ.
//   Both operands need "8 contiguous npq at fixed (k | rsc)" fragments,
//   i.e. npq-innermost LDS images — built by transposed staging
//   (dyT [k][pix], xcolT [c][pix]) like attention's V^T.
//   Block = [64 K] x [32 rsc] per tap-uniform slice; 4 waves each
//   16k x 32rsc; contraction chunks of 64 pixels; fp32 atomicAdd into a
//   dW workspace (gridDim.z = pixel splits) + finalize cast.
//
// bwd-data: dx[n,h,w,c] = sum_{r,s,k} dy[n, (h+pad-r)/st, (w+pad-s)/st, k]
//   * w[k,r,s,c]  (terms only where the division is exact & in range).
//   Mirror of the forward: A = shifted-dy gather (k-contiguous NHWC rows,
//   zero outside / stride-parity mismatch), B = transposed weight image
//   wT[c][k-slice] staged per tap. Block = [128 dx pixels] x [64 c].

#include "dcr_common.h"

namespace dcr_conv_bwd {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

using bf16_t = __hip_bfloat16;

#define PITCH72 72  // npq-innermost rows: 16-lane b128 groups, 16 banks

// ==========================================================================
// bwd-weight
// grid: (K/64, rsc_total/32, pixel_splits); block 256 (4 waves).
// dw_ws: fp32 [K * R*S*C] zero-initialized workspace.
// ==========================================================================
__global__ __launch_bounds__(256)
void conv_bwd_weight_kernel(const bf16_t* __restrict__ dy, const bf16_t* __restrict__ x,
                            float* __restrict__ dw_ws, int Nb, int Hin, int Win,
                            int C, int K, int P, int Q, int R, int S,
                            int stride, int pad, int splits) {
  __shared__ short sDyT[64 * PITCH72];   // [k][pixel]
  __shared__ short sXT[32 * PITCH72];    // [c-of-slice][pixel]

  const int k0 = blockIdx.x * 64;
  const int rsc0 = blockIdx.y * 32;      // tap-uniform slice (C % 32 == 0)
  const int tap = rsc0 / C;
  const int r = tap / S;
  const int s = tap % S;
  const int c0 = rsc0 - tap * C;
  const long NPQ = (long)Nb * P * Q;
  const int rsc_total = R * S * C;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int kgrp = lane >> 4;
  const int wk0 = wid * 16;              // wave's 16 k rows

  f32x4_t acc[2];                        // 16k x 32rsc = 2 n-subtiles
  acc[0] = {0.f, 0.f, 0.f, 0.f};
  acc[1] = {0.f, 0.f, 0.f, 0.f};

  // pixel range of this z-split
  const long chunk_pix = ((NPQ + splits - 1) / splits + 63) / 64 * 64;
  const long pix_begin = (long)blockIdx.z * chunk_pix;
  const long pix_end = min(NPQ, pix_begin + chunk_pix);

  // staging: dyT — thread t reads pixel (t>>2), 16 k at ((t&3)*16),
  // writes transposed; xT — thread t reads pixel (t>>2), c-slice halves.
  const int st_pix = threadIdx.x >> 2;
  const int st_q4 = threadIdx.x & 3;

  for (long pix0 = pix_begin; pix0 < pix_end; pix0 += 64) {
    __syncthreads();
    {
      const long m = pix0 + st_pix;
      int n_ = 0, p_ = 0, q_ = 0;
      bool mok = m < NPQ;
      if (mok) {
        n_ = (int)(m / (P * Q));
        int pq = (int)(m % (P * Q));
        p_ = pq / Q;
        q_ = pq % Q;
      }
      // dyT: 16 k values (k0 + st_q4*16 ..) of this pixel
      {
        short v[16];
        if (mok) {
          const uint4* p4 = reinterpret_cast<const uint4*>(
              dy + m * K + k0 + st_q4 * 16);
          *reinterpret_cast<uint4*>(v) = p4[0];
          *reinterpret_cast<uint4*>(v + 8) = p4[1];
        } else {
#pragma unroll
          for (int j = 0; j < 16; ++j) v[j] = 0;
        }
#pragma unroll
        for (int j = 0; j < 16; ++j)
          sDyT[(st_q4 * 16 + j) * PITCH72 + st_pix] = v[j];
      }
      // xT: threads with st_q4 < 2 handle the two 16-channel halves
      if (st_q4 < 2) {
        short v[16];
        const int hi = p_ * stride + r - pad;
        const int wi = q_ * stride + s - pad;
        if (mok && hi >= 0 && hi < Hin && wi >= 0 && wi < Win) {
          const uint4* p4 = reinterpret_cast<const uint4*>(
              x + (((long)n_ * Hin + hi) * Win + wi) * C + c0 + st_q4 * 16);
          *reinterpret_cast<uint4*>(v) = p4[0];
          *reinterpret_cast<uint4*>(v + 8) = p4[1];
        } else {
#pragma unroll
          for (int j = 0; j < 16; ++j) v[j] = 0;
        }
#pragma unroll
        for (int j = 0; j < 16; ++j)
          sXT[(st_q4 * 16 + j) * PITCH72 + st_pix] = v[j];
      }
    }
    __syncthreads();

    // contraction: two 32-pixel chunks
#pragma unroll
    for (int cc = 0; cc < 2; ++cc) {
      bf16x8 af = *reinterpret_cast<const bf16x8*>(
          sDyT + (wk0 + l16) * PITCH72 + cc * 32 + kgrp * 8);
#pragma unroll
      for (int ns = 0; ns < 2; ++ns) {
        bf16x8 bf = *reinterpret_cast<const bf16x8*>(
            sXT + (ns * 16 + l16) * PITCH72 + cc * 32 + kgrp * 8);
        acc[ns] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc[ns],
                                                          0, 0, 0);
      }
    }
  }

  // C layout: row = k (within wave 16), col = rsc (within 32)
#pragma unroll
  for (int ns = 0; ns < 2; ++ns) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const int k = k0 + wk0 + kgrp * 4 + rr;
      const int rsc = rsc0 + ns * 16 + l16;
      if (k < K)
        atomicAdd(&dw_ws[(long)k * rsc_total + rsc], acc[ns][rr]);
    }
  }
}

__global__ void bwdw_finalize_kernel(const float* __restrict__ ws,
                                     bf16_t* __restrict__ dW, long total) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x)
    dW[i] = __float2bfloat16(ws[i]);
}

// db[k] = sum over output pixels of dy[m][k] (NHWC: k innermost, so
// consecutive lanes read consecutive channels — coalesced). Replaces the
// ~3 ms/step of aten bias-grad reductions once the native bwd dispatches.
// grid (ceil(K/256), row_splits); fp32 atomics into a zeroed db.
__global__ void conv_bias_grad_kernel(const bf16_t* __restrict__ dy,
                                      float* __restrict__ db, long NPQ, int K,
                                      long rows_per_block) {
  const int k = blockIdx.x * blockDim.x + threadIdx.x;
  if (k >= K) return;
  const long r0 = (long)blockIdx.y * rows_per_block;
  const long r1 = min(NPQ, r0 + rows_per_block);
  float s = 0.f;
  for (long m = r0; m < r1; ++m) s += dcr::to_f32<bf16_t>(dy[m * K + k]);
  atomicAdd(&db[k], s);
}

// ==========================================================================
// bwd-data
// grid: (ceil(NHW/128), C/64); block 256 (4 waves as 2x2: 64 pix x 32 c
// per wave... here: wave = 64x64 like fwd v2 but BN=64 -> 2x2 of 64x32).
// Simpler: block [128 dx pixels][64 c]; 4 waves 2x2 each 64pix x 32c.
// K contraction in 32-chunks; 9 (or 1) taps accumulated sequentially.
// ==========================================================================
__global__ __launch_bounds__(256)
void conv_bwd_data_kernel(const bf16_t* __restrict__ dy, const bf16_t* __restrict__ w,
                          bf16_t* __restrict__ dx, int Nb, int Hin, int Win,
                          int C, int K, int P, int Q, int R, int S,
                          int stride, int pad) {
  __shared__ short sDy[128 * 40];        // [pixel][32 k-slice], pitch 40
  __shared__ short sWT[64 * 40];         // [c][32 k-slice], pitch 40

  const long m0 = (long)blockIdx.x * 128;         // dx pixel base
  const int cb0 = blockIdx.y * 64;                // dx channel base
  const long NHW = (long)Nb * Hin * Win;
  const int rsc_total = R * S * C;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int kgrp = lane >> 4;
  const int wr = (wid >> 1) * 64;        // wave pixel base (0|64)
  const int wc = (wid & 1) * 32;         // wave channel base (0|32)

  f32x4_t acc[4][2];                     // 64 pix (4 m-sub) x 32 c (2 n-sub)
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    acc[i][0] = {0.f, 0.f, 0.f, 0.f};
    acc[i][1] = {0.f, 0.f, 0.f, 0.f};
  }

  // this thread's staging pixel for dy (2 threads/row, 16 k each)
  const int st_row = threadIdx.x >> 1;   // 0..127
  const int st_h16 = (threadIdx.x & 1) * 16;
  const long st_m = m0 + st_row;
  int n_ = 0, h_ = 0, w_ = 0;
  if (st_m < NHW) {
    n_ = (int)(st_m / (Hin * Win));
    int hw = (int)(st_m % (Hin * Win));
    h_ = hw / Win;
    w_ = hw % Win;
  }

  for (int tap = 0; tap < R * S; ++tap) {
    const int r = tap / S;
    const int s = tap % S;
    // dy position feeding dx(h,w) through tap (r,s):
    //   p*st = h + pad - r  (exact division required)
    const int hnum = h_ + pad - r;
    const int wnum = w_ + pad - s;
    const bool par_ok = (hnum % stride == 0) && (wnum % stride == 0) &&
                        hnum >= 0 && wnum >= 0;
    const int p_ = par_ok ? hnum / stride : 0;
    const int q_ = par_ok ? wnum / stride : 0;
    const bool dy_ok = st_m < NHW && par_ok && p_ < P && q_ < Q;
    const long dy_base = ((long)n_ * P + p_) * Q + q_;

    for (int kk0 = 0; kk0 < K; kk0 += 32) {
      __syncthreads();
      {
        // dy tile: [pixel][32 k]
        uint4 a0 = make_uint4(0, 0, 0, 0), a1 = a0;
        if (dy_ok) {
          const uint4* p4 = reinterpret_cast<const uint4*>(
              dy + dy_base * K + kk0 + st_h16);
          a0 = p4[0];
          a1 = p4[1];
        }
        // each thread covers 16 of the 32 k: write its half
        uint4* d = reinterpret_cast<uint4*>(sDy + st_row * 40 + st_h16);
        d[0] = a0;
        d[1] = a1;

        // wT tile: [c][32 k] — transposed gather from w[k][rsc];
        // threads 0..127: c = t>>1, k-half = (t&1)*16
        if (threadIdx.x < 128) {
          const int c = cb0 + (threadIdx.x >> 1);
          const int kh = (threadIdx.x & 1) * 16;
          short v[16];
          if (c < C) {
#pragma unroll
            for (int j = 0; j < 16; ++j)
              v[j] = *reinterpret_cast<const short*>(
                  w + (long)(kk0 + kh + j) * rsc_total + tap * C + c);
          } else {
#pragma unroll
            for (int j = 0; j < 16; ++j) v[j] = 0;
          }
#pragma unroll
          for (int j = 0; j < 16; ++j)
            sWT[(threadIdx.x >> 1) * 40 + kh + j] = v[j];
        }
      }
      __syncthreads();

      bf16x8 af[4];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        af[i] = *reinterpret_cast<const bf16x8*>(
            sDy + (wr + i * 16 + l16) * 40 + kgrp * 8);
#pragma unroll
      for (int ns = 0; ns < 2; ++ns) {
        bf16x8 bf = *reinterpret_cast<const bf16x8*>(
            sWT + (wc + ns * 16 + l16) * 40 + kgrp * 8);
#pragma unroll
        for (int i = 0; i < 4; ++i)
          acc[i][ns] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bf,
                                                               acc[i][ns],
                                                               0, 0, 0);
      }
    }
  }

#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const long m = m0 + wr + i * 16 + kgrp * 4 + rr;
      if (m >= NHW) continue;
#pragma unroll
      for (int ns = 0; ns < 2; ++ns) {
        const int c = cb0 + wc + ns * 16 + l16;
        if (c < C)
          dx[m * C + c] = __float2bfloat16(acc[i][ns][rr]);
      }
    }
  }
}

}  // namespace dcr_conv_bwd

#include "dcr_launchers.h"

namespace dcr {

void conv_bwd_weight_launch(const void* dy, const void* x, float* dw_ws,
                            int Nb, int Hin, int Win, int C, int K, int P,
                            int Q, int R, int S, int stride, int pad,
                            int splits, void* dW_out, hipStream_t st) {
  dim3 grid((unsigned)(K / 64), (unsigned)(R * S * C / 32), (unsigned)splits),
      block(256);
  hipLaunchKernelGGL(dcr_conv_bwd::conv_bwd_weight_kernel, grid, block, 0, st,
                     (const dcr_conv_bwd::bf16_t*)dy,
                     (const dcr_conv_bwd::bf16_t*)x, dw_ws, Nb, Hin, Win, C,
                     K, P, Q, R, S, stride, pad, splits);
  long total = (long)K * R * S * C;
  long b = (total + 255) / 256;
  if (b > 8192) b = 8192;
  hipLaunchKernelGGL(dcr_conv_bwd::bwdw_finalize_kernel, dim3((unsigned)b),
                     dim3(256), 0, st, dw_ws,
                     (dcr_conv_bwd::bf16_t*)dW_out, total);
}

void conv_bias_grad_launch(const void* dy, float* db, long NPQ, int K,
                           hipStream_t st) {
  long rows_per_block = 4096;
  unsigned ysplit = (unsigned)((NPQ + rows_per_block - 1) / rows_per_block);
  if (ysplit > 1024) {  // keep the grid bounded for huge NPQ
    rows_per_block = (NPQ + 1023) / 1024;
    ysplit = (unsigned)((NPQ + rows_per_block - 1) / rows_per_block);
  }
  dim3 grid((unsigned)((K + 255) / 256), ysplit), block(256);
  hipLaunchKernelGGL(dcr_conv_bwd::conv_bias_grad_kernel, grid, block, 0, st,
                     (const dcr_conv_bwd::bf16_t*)dy, db, NPQ, K,
                     rows_per_block);
}

void conv_bwd_data_launch(const void* dy, const void* w, void* dx, int Nb,
                          int Hin, int Win, int C, int K, int P, int Q, int R,
                          int S, int stride, int pad, hipStream_t st) {
  long NHW = (long)Nb * Hin * Win;
  dim3 grid((unsigned)((NHW + 127) / 128), (unsigned)(C / 64)), block(256);
  hipLaunchKernelGGL(dcr_conv_bwd::conv_bwd_data_kernel, grid, block, 0, st,
                     (const dcr_conv_bwd::bf16_t*)dy,
                     (const dcr_conv_bwd::bf16_t*)w,
                     (dcr_conv_bwd::bf16_t*)dx, Nb, Hin, Win, C, K, P, Q, R,
                     S, stride, pad);
}

}  // namespace dcr
