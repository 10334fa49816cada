// Implicit-GEMM convolution forward (NHWC, bf16, MFMA) for MI355X.
//
// SURVEY.md §7 "hard parts" item 2: hand-written implicit-GEMM conv for
// the SD UNet/VAE shapes. GEMM view of out[n,p,q,k] = sum_{r,s,c}
// x[n, p*stride+r-pad, q*stride+s-pad, c] * w[k,r,s,c]:
//   M = N*P*Q output pixels, N = K output channels, K-dim = R*S*C.
// Same fragment discipline as attention.hip (v_mfma_f32_16x16x32_bf16,
// "8 contiguous contraction elements at row lane&15"):
//   * A (pixels)  : on-the-fly im2col — a 32-wide rsc slice with C%32==0
//     never straddles an (r,s) tap, so each pixel's slice is 32
//     CONTIGUOUS channels of one input row; staged to LDS (pitch 40).
//   * B (weights) : [K,R,S,C] channels_last weight memory is contiguous
//     in rsc for fixed k — fragments read DIRECTLY from global (the
//     weight tile stays hot in L2 across the many pixel blocks).
// Block = 4 waves = 64 pixels x 64 out-channels; grid (pixels/64, K/64, 1).
//
// Constraints (dispatcher falls back to MIOpen otherwise): bf16,
// channels_last, C % 32 == 0, K % 64 == 0, kernel 3x3(pad 1) or
// 1x1(pad 0), stride 1 or 2. Opt-in via DCR_NATIVE_CONV=1 (fwd only —
// autograd uses MIOpen for backward).

#include "dcr_common.h"

namespace dcr_conv {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

using bf16_t = __hip_bfloat16;

#define CPITCH 40  // LDS row pitch (bf16) for the 32-wide A tile: 10 dwords
                   // -> 16-lane ds_read_b128 groups land on 16 distinct banks

__global__ __launch_bounds__(256)
void conv_nhwc_fwd_kernel(const bf16_t* __restrict__ x, const bf16_t* __restrict__ w,
                          const float* __restrict__ bias, bf16_t* __restrict__ y,
                          int Nb, int Hin, int Win, int C, int K, int P, int Q,
                          int R, int S, int stride, int pad) {
  __shared__ short sA[64 * CPITCH];

  const int m0 = blockIdx.x * 64;          // first output pixel of this block
  const int k0 = blockIdx.y * 64;          // first output channel
  const long NPQ = (long)Nb * P * Q;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int kgrp = lane >> 4;
  const int wrow0 = wid * 16;              // wave's 16 pixels

  f32x4_t acc[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) acc[i] = {0.f, 0.f, 0.f, 0.f};

  // this thread's staging pixel: 4 threads per pixel row, 8 bf16 each
  const int st_pix = threadIdx.x >> 2;     // 0..63
  const int st_c8 = (threadIdx.x & 3) * 8; // 0..24 within the 32-slice
  long st_m = m0 + st_pix;
  int st_n = 0, st_p = 0, st_q = 0;
  if (st_m < NPQ) {
    st_n = (int)(st_m / (P * Q));
    int pq = (int)(st_m % (P * Q));
    st_p = pq / Q;
    st_q = pq % Q;
  }

  const int rsc_total = R * S * C;
  for (int rsc0 = 0; rsc0 < rsc_total; rsc0 += 32) {
    // decode the tap for this 32-slice (C % 32 == 0: tap is slice-uniform)
    const int tap = rsc0 / C;
    const int r = tap / S;
    const int s = tap % S;
    const int c0 = rsc0 - tap * C;

    __syncthreads();
    {
      // stage A: pixel st_pix, channels [c0+st_c8, +8) of tap (r, s)
      uint4 v = make_uint4(0, 0, 0, 0);
      const int hi = st_p * stride + r - pad;
      const int wi = st_q * stride + s - pad;
      if (st_m < NPQ && hi >= 0 && hi < Hin && wi >= 0 && wi < Win) {
        const bf16_t* src = x + (((long)st_n * Hin + hi) * Win + wi) * C + c0 + st_c8;
        v = *reinterpret_cast<const uint4*>(src);
      }
      *reinterpret_cast<uint4*>(sA + st_pix * CPITCH + st_c8) = v;
    }
    __syncthreads();

    // A fragment: pixel row (wrow0 + l16), k-elems kgrp*8..+8 of the slice
    bf16x8 af = *reinterpret_cast<const bf16x8*>(
        sA + (wrow0 + l16) * CPITCH + kgrp * 8);
    // B fragments straight from global: w[k0 + ns*16 + l16][rsc0 + kgrp*8]
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
      const bf16_t* wp = w + (long)(k0 + ns * 16 + l16) * rsc_total + rsc0 + kgrp * 8;
      bf16x8 bf = *reinterpret_cast<const bf16x8*>(wp);
      acc[ns] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af, bf, acc[ns], 0, 0, 0);
    }
  }

  // epilogue: += bias, store [pixel][k] (NHWC output: k contiguous)
#pragma unroll
  for (int rr = 0; rr < 4; ++rr) {
    const long m = m0 + wrow0 + kgrp * 4 + rr;
    if (m >= NPQ) continue;
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
      const int k = k0 + ns * 16 + l16;
      float v = acc[ns][rr] + (bias ? bias[k] : 0.f);
      y[m * K + k] = __float2bfloat16(v);
    }
  }
}

}  // namespace dcr_conv

#include "dcr_launchers.h"

namespace dcr {

void conv_nhwc_fwd_launch(const void* x, const void* w, const float* bias,
                          void* y, int Nb, int Hin, int Win, int C, int K,
                          int P, int Q, int R, int S, int stride, int pad,
                          hipStream_t st) {
  long NPQ = (long)Nb * P * Q;
  dim3 grid((unsigned)((NPQ + 63) / 64), (unsigned)(K / 64)), block(256);
  hipLaunchKernelGGL(dcr_conv::conv_nhwc_fwd_kernel, grid, block, 0, st,
                     (const dcr_conv::bf16_t*)x, (const dcr_conv::bf16_t*)w,
                     bias, (dcr_conv::bf16_t*)y, Nb, Hin, Win, C, K, P, Q, R,
                     S, stride, pad);
}

}  // namespace dcr

// ===========================================================================
// v2: 128x128 tile, LDS-staged A (im2col gather) and B (weights), BK=32.
// 4 waves as 2x2, each computing a 64x64 sub-tile (acc[4][4] f32x4).
// Still single-buffered (2 barriers per K-step) — the guide's "step-2"
// structure; double-buffering + split-K are the round-2 follow-ups.
// ===========================================================================
namespace dcr_conv {

// splitz > 1: blockIdx.z covers a slice of the rsc steps; fp32 partials
// are atomically accumulated into ws[NPQ*K] and a finalize kernel adds
// bias + casts (grid starvation fix for the 8x8/16x16 shapes).
// res: optional residual [NPQ, K] (NHWC tensor) added in the epilogue
// (ResnetBlock2D's `x + h`); temb: optional per-(n,k) bf16 bias [N, K]
// (the time-embedding projection broadcast) — both fuse away a full
// elementwise pass over the output tensor.
template <int BK>
__global__ __launch_bounds__(256)
void conv_nhwc_fwd_v2_kernel(const bf16_t* __restrict__ x, const bf16_t* __restrict__ w,
                             const float* __restrict__ bias, bf16_t* __restrict__ y,
                             float* __restrict__ ws, int splitz,
                             const bf16_t* __restrict__ res,
                             const bf16_t* __restrict__ temb,
                             int Nb, int Hin, int Win, int C, int K, int P, int Q,
                             int R, int S, int stride, int pad) {
  constexpr int PITCH2 = BK + 8;           // 16-lane b128 groups: 16 banks
  __shared__ short sA[128 * PITCH2];
  __shared__ short sB[128 * PITCH2];

  const long m0 = (long)blockIdx.x * 128;
  const int k0 = blockIdx.y * 128;
  const long NPQ = (long)Nb * P * Q;
  const int rsc_total = R * S * C;

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int kgrp = lane >> 4;
  const int wr = (wid >> 1) * 64;          // wave pixel base within tile
  const int wc = (wid & 1) * 64;           // wave k base within tile

  f32x4_t acc[4][4];
#pragma unroll
  for (int i = 0; i < 4; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  // staging: 2 threads per row, each covers BK/2 contiguous bf16
  const int st_row = threadIdx.x >> 1;     // 0..127
  const int st_chalf = (threadIdx.x & 1) * (BK / 2);  // half-row column base
  long st_m = m0 + st_row;
  int st_n = 0, st_p = 0, st_q = 0;
  if (st_m < NPQ) {
    st_n = (int)(st_m / (P * Q));
    int pq = (int)(st_m % (P * Q));
    st_p = pq / Q;
    st_q = pq % Q;
  }
  const long wrow = (long)(k0 + st_row) * rsc_total;  // B source row base

  const int nsteps = rsc_total / BK;
  const int spz = (nsteps + splitz - 1) / splitz;
  const int step0 = blockIdx.z * spz;
  const int step1 = min(nsteps, step0 + spz);
  constexpr int NCH = BK / 16;             // 8-elem uint4 chunks per half-row
  constexpr int HALF = BK / 2;

  for (int step = step0; step < step1; ++step) {
    const int rsc0 = step * BK;
    const int tap = rsc0 / C;              // C % BK == 0: tap slice-uniform
    const int r = tap / S;
    const int s = tap % S;
    const int c0 = rsc0 - tap * C;

    __syncthreads();
    {
      uint4 av[NCH], bv[NCH];
#pragma unroll
      for (int t = 0; t < NCH; ++t) av[t] = make_uint4(0, 0, 0, 0);
#pragma unroll
      for (int t = 0; t < NCH; ++t) bv[t] = make_uint4(0, 0, 0, 0);
      const int hi = st_p * stride + r - pad;
      const int wi = st_q * stride + s - pad;
      if (st_m < NPQ && hi >= 0 && hi < Hin && wi >= 0 && wi < Win) {
        const uint4* p4 = reinterpret_cast<const uint4*>(
            x + (((long)st_n * Hin + hi) * Win + wi) * C + c0 + st_chalf);
#pragma unroll
        for (int t = 0; t < NCH; ++t) av[t] = p4[t];
      }
      if (k0 + st_row < K) {
        const uint4* p4 = reinterpret_cast<const uint4*>(w + wrow + rsc0 + st_chalf);
#pragma unroll
        for (int t = 0; t < NCH; ++t) bv[t] = p4[t];
      }
      uint4* d = reinterpret_cast<uint4*>(sA + st_row * PITCH2 + st_chalf);
      uint4* db = reinterpret_cast<uint4*>(sB + st_row * PITCH2 + st_chalf);
#pragma unroll
      for (int t = 0; t < NCH; ++t) d[t] = av[t];
#pragma unroll
      for (int t = 0; t < NCH; ++t) db[t] = bv[t];
    }
    __syncthreads();

#pragma unroll
    for (int kk = 0; kk < BK / 32; ++kk) {
      bf16x8 af[4], bf[4];
#pragma unroll
      for (int i = 0; i < 4; ++i)
        af[i] = *reinterpret_cast<const bf16x8*>(
            sA + (wr + i * 16 + l16) * PITCH2 + kk * 32 + kgrp * 8);
#pragma unroll
      for (int j = 0; j < 4; ++j)
        bf[j] = *reinterpret_cast<const bf16x8*>(
            sB + (wc + j * 16 + l16) * PITCH2 + kk * 32 + kgrp * 8);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bf[j],
                                                              acc[i][j], 0, 0, 0);
    }
  }

  const int PQ = P * Q;
#pragma unroll
  for (int i = 0; i < 4; ++i) {
#pragma unroll
    for (int rr = 0; rr < 4; ++rr) {
      const long m = m0 + wr + i * 16 + kgrp * 4 + rr;
      if (m >= NPQ) continue;
#pragma unroll
      for (int j = 0; j < 4; ++j) {
        const int k = k0 + wc + j * 16 + l16;
        if (k >= K) continue;
        if (splitz > 1) {
          atomicAdd(&ws[m * K + k], acc[i][j][rr]);
        } else {
          float v = acc[i][j][rr] + (bias ? bias[k] : 0.f);
          if (res) v += __bfloat162float(res[m * K + k]);
          if (temb) v += __bfloat162float(temb[(m / PQ) * K + k]);
          y[m * K + k] = __float2bfloat16(v);
        }
      }
    }
  }
}

__global__ void conv_splitk_finalize_kernel(const float* __restrict__ ws,
                                            const float* __restrict__ bias,
                                            const bf16_t* __restrict__ res,
                                            const bf16_t* __restrict__ temb,
                                            bf16_t* __restrict__ y, long total,
                                            int K, int PQ) {
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < total;
       i += (long)gridDim.x * blockDim.x) {
    float v = ws[i] + (bias ? bias[i % K] : 0.f);
    if (res) v += __bfloat162float(res[i]);
    if (temb) v += __bfloat162float(temb[(i / ((long)K * PQ)) * K + i % K]);
    y[i] = __float2bfloat16(v);
  }
}

}  // namespace dcr_conv

namespace dcr {

void conv_nhwc_fwd_v2_launch(const void* x, const void* w, const float* bias,
                             void* y, float* ws, int splitz, const void* res,
                             const void* temb, int Nb, int Hin, int Win, int C,
                             int K, int P, int Q, int R, int S, int stride,
                             int pad, hipStream_t st) {
  long NPQ = (long)Nb * P * Q;
  dim3 grid((unsigned)((NPQ + 127) / 128), (unsigned)((K + 127) / 128),
            (unsigned)splitz),
      block(256);
  const auto* rp = (const dcr_conv::bf16_t*)res;
  const auto* tp = (const dcr_conv::bf16_t*)temb;
  if (C % 64 == 0)
    hipLaunchKernelGGL((dcr_conv::conv_nhwc_fwd_v2_kernel<64>), grid, block, 0,
                       st, (const dcr_conv::bf16_t*)x,
                       (const dcr_conv::bf16_t*)w, bias, (dcr_conv::bf16_t*)y,
                       ws, splitz, rp, tp, Nb, Hin, Win, C, K, P, Q, R, S,
                       stride, pad);
  else
    hipLaunchKernelGGL((dcr_conv::conv_nhwc_fwd_v2_kernel<32>), grid, block, 0,
                       st, (const dcr_conv::bf16_t*)x,
                       (const dcr_conv::bf16_t*)w, bias, (dcr_conv::bf16_t*)y,
                       ws, splitz, rp, tp, Nb, Hin, Win, C, K, P, Q, R, S,
                       stride, pad);
  if (splitz > 1) {
    long total = NPQ * K;
    long b = (total / 4 + 255) / 256;
    if (b > 8192) b = 8192;
    hipLaunchKernelGGL(dcr_conv::conv_splitk_finalize_kernel,
                       dim3((unsigned)b), dim3(256), 0, st, ws, bias, rp, tp,
                       (dcr_conv::bf16_t*)y, total, K, P * Q);
  }
}

}  // namespace dcr
