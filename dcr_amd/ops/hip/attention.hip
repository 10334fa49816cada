// Flash attention (forward + backward) for MI355X (gfx950), bf16, D=64.
//
// The hand-written CDNA4 FMHA required by the north star: LDS-staged
// K/V tiles, MFMA (v_mfma_f32_16x16x32_bf16) for QK^T / PV and the
// backward GEMMs, online softmax with fp32 row stats, wave64 quarter-
// group row reductions. Replaces torch SDPA (AOTriton) for SD-2.1
// attention (head_dim 64: latent self-attn L in {64,256,1024,4096},
// text cross-attn Lk=77 — arbitrary lengths via tile masking) and CLIP
// text (causal L=77). Reference ops covered: SURVEY.md §2.4.A FMHA rows.
//
// Fragment conventions (gfx950 v_mfma_f32_16x16x32_bf16, K-contiguous;
// verified on hardware by the mfma_probe test in tests/test_ops_gpu.py):
//   A[m][k]  : lane l holds row  m=l&15, k = (l>>4)*8 + j   (8 bf16)
//   B[k][n]  : lane l holds col  n=l&15, k = (l>>4)*8 + j
//   C/D[m][n]: lane l holds col  n=l&15, rows m=(l>>4)*4 + r (4 fp32)
// Every LDS fragment read is "8 contiguous contraction elements at row
// l&15", so tiles are stored with the contraction dim innermost; row
// pitch 72 bf16 (144 B) puts the 16 rows of a ds_read_b128 lane group on
// 16 distinct banks (36*r mod 64 is a 16-cycle) — conflict-free without
// an XOR swizzle.
//
// Backward is the FlashAttention-2 split: attn_bwd_dkdv accumulates
// dK/dV per key tile (loops q tiles), attn_bwd_dq accumulates dQ per
// query tile (loops key tiles); both recompute P from (Q, K, LSE);
// delta = rowsum(dO*O) precomputed by attn_bwd_delta.

#include "dcr_common.h"

namespace dcr_attn {

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;

#define TILE 64
#define PITCH 72
#define DHEAD 64

using bf16_t = __hip_bfloat16;

__device__ __forceinline__ short f2bf_rne(float f) {
  __hip_bfloat16 h = __float2bfloat16(f);
  return *reinterpret_cast<short*>(&h);
}

// cooperative tile load (256 threads): dst[row][dim], zero-padded rows.
// rs = element stride between consecutive sequence rows (64 for [BH,L,D]
// packed, H*64 for the transpose-free [B,L,H,D] layout).
__device__ __forceinline__ void load_tile(const bf16_t* __restrict__ src,
                                          long rs, int valid_rows,
                                          short* __restrict__ dst) {
  const int t = threadIdx.x;
  const int row = t >> 2;
  const int col = (t & 3) * 16;
  uint4 a = make_uint4(0, 0, 0, 0), b = a;
  if (row < valid_rows) {
    const uint4* p = reinterpret_cast<const uint4*>(src + (long)row * rs + col);
    a = p[0];
    b = p[1];
  }
  uint4* d = reinterpret_cast<uint4*>(dst + row * PITCH + col);
  d[0] = a;
  d[1] = b;
}

// transposed: dst[dim][row] = src[row][dim]
__device__ __forceinline__ void load_tile_T(const bf16_t* __restrict__ src,
                                            long rs, int valid_rows,
                                            short* __restrict__ dst) {
  const int t = threadIdx.x;
  const int row = t >> 2;
  const int col0 = (t & 3) * 16;
  short v[16];
  if (row < valid_rows) {
    const uint4* p = reinterpret_cast<const uint4*>(src + (long)row * rs + col0);
    *reinterpret_cast<uint4*>(v) = p[0];
    *reinterpret_cast<uint4*>(v + 8) = p[1];
  } else {
#pragma unroll
    for (int j = 0; j < 16; ++j) v[j] = 0;
  }
#pragma unroll
  for (int j = 0; j < 16; ++j) dst[(col0 + j) * PITCH + row] = v[j];
}

__device__ __forceinline__ bf16x8 frag(const short* lds, int row, int koff) {
  return *reinterpret_cast<const bf16x8*>(lds + row * PITCH + koff);
}

// 16-lane (quarter-wave) reductions — rows are spread over lanes l^1..l^8
__device__ __forceinline__ float qmax(float v) {
#pragma unroll
  for (int m = 8; m >= 1; m >>= 1) v = fmaxf(v, __shfl_xor(v, m, 64));
  return v;
}
__device__ __forceinline__ float qsum(float v) {
#pragma unroll
  for (int m = 8; m >= 1; m >>= 1) v += __shfl_xor(v, m, 64);
  return v;
}

// ==========================================================================
// Forward. grid (ceil(Lq/64), BH), block 256.
// ==========================================================================
__global__ __launch_bounds__(256)
void attn_fwd_kernel(const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
                     const bf16_t* __restrict__ v, bf16_t* __restrict__ o,
                     float* __restrict__ lse, int Lq, int Lk, int H,
                     float scale, int causal) {
  __shared__ short sQ[TILE * PITCH];
  __shared__ short sK[TILE * PITCH];
  __shared__ short sVT[TILE * PITCH];   // [dim][key]
  __shared__ short sP[TILE * PITCH];    // [qrow][key]

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int q0 = blockIdx.x * TILE;
  const long qrs = (long)H * DHEAD;       // row stride in the [B,L,H,D] walk
  const bf16_t* qp = q + (((long)b * Lq + q0) * H + h) * DHEAD;
  const bf16_t* kp = k + ((long)b * Lk * H + h) * DHEAD;
  const bf16_t* vp = v + ((long)b * Lk * H + h) * DHEAD;
  bf16_t* op = o + ((long)b * Lq * H + h) * DHEAD;

  load_tile(qp, qrs, Lq - q0, sQ);

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int kgrp = lane >> 4;
  const int wrow0 = wid * 16;

  float row_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  float row_sum[4] = {0.f, 0.f, 0.f, 0.f};
  f32x4_t acc_o[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) acc_o[i] = {0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? min(Lk, q0 + TILE) : Lk;

  for (int kv0 = 0; kv0 < kv_end; kv0 += TILE) {
    __syncthreads();
    load_tile(kp + (long)kv0 * qrs, qrs, Lk - kv0, sK);
    load_tile_T(vp + (long)kv0 * qrs, qrs, Lk - kv0, sVT);
    __syncthreads();

    // S = scale * Q K^T  (wave: 16 q-rows x 64 keys)
    bf16x8 qf0 = frag(sQ, wrow0 + l16, kgrp * 8);
    bf16x8 qf1 = frag(sQ, wrow0 + l16, kgrp * 8 + 32);
    f32x4_t s_frag[4];
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
      f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          qf0, frag(sK, ns * 16 + l16, kgrp * 8), acc, 0, 0, 0);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          qf1, frag(sK, ns * 16 + l16, kgrp * 8 + 32), acc, 0, 0, 0);
      s_frag[ns] = acc;
    }

    // mask + online softmax
    float tile_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
      const int key = kv0 + ns * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float sv = s_frag[ns][r] * scale;
        const int qrow = q0 + wrow0 + kgrp * 4 + r;
        if (key >= Lk || (causal && key > qrow)) sv = -1e30f;
        s_frag[ns][r] = sv;
        tile_max[r] = fmaxf(tile_max[r], sv);
      }
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float tm = qmax(tile_max[r]);
      float mnew = fmaxf(row_max[r], tm);
      alpha[r] = (mnew <= -1e29f) ? 1.f : __expf(row_max[r] - mnew);
      row_max[r] = mnew;
    }

    float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float pv = (s_frag[ns][r] <= -1e29f)
                       ? 0.f : __expf(s_frag[ns][r] - row_max[r]);
        psum[r] += pv;
        sP[(wrow0 + kgrp * 4 + r) * PITCH + ns * 16 + l16] = f2bf_rne(pv);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) row_sum[r] = row_sum[r] * alpha[r] + qsum(psum[r]);

#pragma unroll
    for (int ds_ = 0; ds_ < 4; ++ds_)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[ds_][r] *= alpha[r];

    // O += P V   (contraction 64 keys, two 32-chunks; same-wave P reuse)
    bf16x8 pf0 = frag(sP, wrow0 + l16, kgrp * 8);
    bf16x8 pf1 = frag(sP, wrow0 + l16, kgrp * 8 + 32);
#pragma unroll
    for (int ds_ = 0; ds_ < 4; ++ds_) {
      acc_o[ds_] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          pf0, frag(sVT, ds_ * 16 + l16, kgrp * 8), acc_o[ds_], 0, 0, 0);
      acc_o[ds_] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          pf1, frag(sVT, ds_ * 16 + l16, kgrp * 8 + 32), acc_o[ds_], 0, 0, 0);
    }
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + wrow0 + kgrp * 4 + r;
    if (qrow >= Lq) continue;
    const float inv = (row_sum[r] > 0.f) ? 1.f / row_sum[r] : 0.f;
#pragma unroll
    for (int ds_ = 0; ds_ < 4; ++ds_)
      op[(long)qrow * qrs + ds_ * 16 + l16] = __float2bfloat16(acc_o[ds_][r] * inv);
    if (l16 == 0 && lse != nullptr)
      lse[(long)bh * Lq + qrow] = row_max[r] + __logf(fmaxf(row_sum[r], 1e-30f));
  }
}

// ==========================================================================
// Backward preprocess: delta[row] = sum_d dO[row,d]*O[row,d]
// ==========================================================================
// ==========================================================================
// Forward v2 (round-2 draft): identical structure + masked-tail MFMA skip.
// For the last key tile (and the whole pass at Lk<=64), every 16-key
// ns-subtile beyond ceil(valid/16) is fully masked: its S values are
// -inf before the exp, so its sP columns are zeros and its PV products
// contribute nothing. Skipping those MFMAs is therefore BIT-EXACT vs v1
// (fp32 accumulation of exact zeros) while saving 6 of 8 S-MFMAs and 4
// of 8 PV-MFMAs on the Lk=77 cross-attention tail tile (13 valid keys).
// NOT dispatched this round — gated tests assert torch.equal vs v1
// (DCR_ATTN_V2=1), A/B in scripts/bench_attention.py.
// ==========================================================================
__global__ __launch_bounds__(256)
void attn_fwd_v2_kernel(const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
                        const bf16_t* __restrict__ v, bf16_t* __restrict__ o,
                        float* __restrict__ lse, int Lq, int Lk, int H,
                        float scale, int causal) {
  __shared__ short sQ[TILE * PITCH];
  __shared__ short sK[TILE * PITCH];
  __shared__ short sVT[TILE * PITCH];
  __shared__ short sP[TILE * PITCH];

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int q0 = blockIdx.x * TILE;
  const long qrs = (long)H * DHEAD;
  const bf16_t* qp = q + (((long)b * Lq + q0) * H + h) * DHEAD;
  const bf16_t* kp = k + ((long)b * Lk * H + h) * DHEAD;
  const bf16_t* vp = v + ((long)b * Lk * H + h) * DHEAD;
  bf16_t* op = o + ((long)b * Lq * H + h) * DHEAD;

  load_tile(qp, qrs, Lq - q0, sQ);

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int kgrp = lane >> 4;
  const int wrow0 = wid * 16;

  float row_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  float row_sum[4] = {0.f, 0.f, 0.f, 0.f};
  f32x4_t acc_o[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) acc_o[i] = {0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? min(Lk, q0 + TILE) : Lk;

  for (int kv0 = 0; kv0 < kv_end; kv0 += TILE) {
    __syncthreads();
    load_tile(kp + (long)kv0 * qrs, qrs, Lk - kv0, sK);
    load_tile_T(vp + (long)kv0 * qrs, qrs, Lk - kv0, sVT);
    __syncthreads();

    // number of 16-key subtiles with any valid key in this tile
    const int kv_valid = min(TILE, Lk - kv0);
    const int nv = (kv_valid + 15) >> 4;

    bf16x8 qf0 = frag(sQ, wrow0 + l16, kgrp * 8);
    bf16x8 qf1 = frag(sQ, wrow0 + l16, kgrp * 8 + 32);
    f32x4_t s_frag[4];
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
      f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
      if (ns < nv) {
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qf0, frag(sK, ns * 16 + l16, kgrp * 8), acc, 0, 0, 0);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            qf1, frag(sK, ns * 16 + l16, kgrp * 8 + 32), acc, 0, 0, 0);
      }
      s_frag[ns] = acc;
    }

    float tile_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
      const int key = kv0 + ns * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float sv = s_frag[ns][r] * scale;
        const int qrow = q0 + wrow0 + kgrp * 4 + r;
        if (key >= Lk || (causal && key > qrow)) sv = -1e30f;
        s_frag[ns][r] = sv;
        tile_max[r] = fmaxf(tile_max[r], sv);
      }
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float tm = qmax(tile_max[r]);
      float mnew = fmaxf(row_max[r], tm);
      alpha[r] = (mnew <= -1e29f) ? 1.f : __expf(row_max[r] - mnew);
      row_max[r] = mnew;
    }

    float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float pv = (s_frag[ns][r] <= -1e29f)
                       ? 0.f : __expf(s_frag[ns][r] - row_max[r]);
        psum[r] += pv;
        sP[(wrow0 + kgrp * 4 + r) * PITCH + ns * 16 + l16] = f2bf_rne(pv);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) row_sum[r] = row_sum[r] * alpha[r] + qsum(psum[r]);

#pragma unroll
    for (int ds_ = 0; ds_ < 4; ++ds_)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[ds_][r] *= alpha[r];

    // PV: the high 32 keys of the tile are all invalid when nv <= 2 —
    // their sP columns are zeros, skip the second MFMA pair entirely
    bf16x8 pf0 = frag(sP, wrow0 + l16, kgrp * 8);
#pragma unroll
    for (int ds_ = 0; ds_ < 4; ++ds_)
      acc_o[ds_] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          pf0, frag(sVT, ds_ * 16 + l16, kgrp * 8), acc_o[ds_], 0, 0, 0);
    if (nv > 2) {
      bf16x8 pf1 = frag(sP, wrow0 + l16, kgrp * 8 + 32);
#pragma unroll
      for (int ds_ = 0; ds_ < 4; ++ds_)
        acc_o[ds_] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            pf1, frag(sVT, ds_ * 16 + l16, kgrp * 8 + 32), acc_o[ds_], 0, 0, 0);
    }
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + wrow0 + kgrp * 4 + r;
    if (qrow >= Lq) continue;
    const float inv = (row_sum[r] > 0.f) ? 1.f / row_sum[r] : 0.f;
#pragma unroll
    for (int ds_ = 0; ds_ < 4; ++ds_)
      op[(long)qrow * qrs + ds_ * 16 + l16] = __float2bfloat16(acc_o[ds_][r] * inv);
    if (l16 == 0 && lse != nullptr)
      lse[(long)bh * Lq + qrow] = row_max[r] + __logf(fmaxf(row_sum[r], 1e-30f));
  }
}

__global__ void attn_bwd_delta_kernel(const bf16_t* __restrict__ dO,
                                      const bf16_t* __restrict__ O,
                                      float* __restrict__ delta, long total_rows,
                                      int Lq, int H) {
  long row = (long)blockIdx.x * blockDim.x + threadIdx.x;  // (b*H + h)*Lq + l
  if (row >= total_rows) return;
  const long bh = row / Lq, l = row % Lq;
  const long b_ = bh / H, h = bh % H;
  const long off = ((b_ * Lq + l) * H + h) * DHEAD;
  const bf16_t* a = dO + off;
  const bf16_t* b = O + off;
  float s = 0.f;
#pragma unroll
  for (int i = 0; i < DHEAD; i += 4) {
    dcr::f32x4 av = dcr::load4<__hip_bfloat16>(a + i);
    dcr::f32x4 bv = dcr::load4<__hip_bfloat16>(b + i);
    s += av.x * bv.x + av.y * bv.y + av.z * bv.z + av.w * bv.w;
  }
  delta[row] = s;
}

// ==========================================================================
// Backward dK/dV. grid (ceil(Lk/64), BH); wave owns 16 keys.
// ==========================================================================
__global__ __launch_bounds__(256)
void attn_bwd_dkdv_kernel(const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
                          const bf16_t* __restrict__ v, const bf16_t* __restrict__ dO,
                          const float* __restrict__ lse, const float* __restrict__ delta,
                          bf16_t* __restrict__ dK, bf16_t* __restrict__ dV,
                          int Lq, int Lk, int H, float scale, int causal) {
  __shared__ short sK[TILE * PITCH];     // [key][dim]
  __shared__ short sV[TILE * PITCH];     // [key][dim]
  __shared__ short sQ[TILE * PITCH];     // [qrow][dim]
  __shared__ short sQT[TILE * PITCH];    // [dim][qrow]
  __shared__ short sdO[TILE * PITCH];    // [qrow][dim]
  __shared__ short sdOT[TILE * PITCH];   // [dim][qrow]
  __shared__ short sPT[TILE * PITCH];    // [key][qrow]: P^T, then dS^T
  __shared__ float sLse[TILE];
  __shared__ float sDelta[TILE];

  const int bh = blockIdx.y;
  const int b_ = bh / H, h = bh % H;
  const int k0 = blockIdx.x * TILE;
  const long rs = (long)H * DHEAD;
  const bf16_t* qp = q + ((long)b_ * Lq * H + h) * DHEAD;
  const bf16_t* kp = k + (((long)b_ * Lk + k0) * H + h) * DHEAD;
  const bf16_t* vp = v + (((long)b_ * Lk + k0) * H + h) * DHEAD;
  const bf16_t* dop = dO + ((long)b_ * Lq * H + h) * DHEAD;
  const float* lsep = lse + (long)bh * Lq;
  const float* delp = delta + (long)bh * Lq;

  load_tile(kp, rs, Lk - k0, sK);
  load_tile(vp, rs, Lk - k0, sV);

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int kgrp = lane >> 4;
  const int wkey0 = wid * 16;

  f32x4_t acc_dk[4], acc_dv[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    acc_dk[i] = {0.f, 0.f, 0.f, 0.f};
    acc_dv[i] = {0.f, 0.f, 0.f, 0.f};
  }

  const int q_start = causal ? k0 : 0;

  for (int q0 = q_start; q0 < Lq; q0 += TILE) {
    __syncthreads();
    load_tile(qp + (long)q0 * rs, rs, Lq - q0, sQ);
    load_tile_T(qp + (long)q0 * rs, rs, Lq - q0, sQT);
    load_tile(dop + (long)q0 * rs, rs, Lq - q0, sdO);
    load_tile_T(dop + (long)q0 * rs, rs, Lq - q0, sdOT);
    if (threadIdx.x < TILE) {
      const int qr = q0 + threadIdx.x;
      sLse[threadIdx.x] = (qr < Lq) ? lsep[qr] : 1e30f;
      sDelta[threadIdx.x] = (qr < Lq) ? delp[qr] : 0.f;
    }
    __syncthreads();

    bf16x8 kf0 = frag(sK, wkey0 + l16, kgrp * 8);
    bf16x8 kf1 = frag(sK, wkey0 + l16, kgrp * 8 + 32);
    bf16x8 vf0 = frag(sV, wkey0 + l16, kgrp * 8);
    bf16x8 vf1 = frag(sV, wkey0 + l16, kgrp * 8 + 32);

    f32x4_t dst_all[4];                  // dS^T frags per query subtile

#pragma unroll
    for (int ms = 0; ms < 4; ++ms) {
      // S^T[key][m] = K Q^T: A=K (k=dim), B=Q rows (k=dim)
      f32x4_t st = {0.f, 0.f, 0.f, 0.f};
      st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          kf0, frag(sQ, ms * 16 + l16, kgrp * 8), st, 0, 0, 0);
      st = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          kf1, frag(sQ, ms * 16 + l16, kgrp * 8 + 32), st, 0, 0, 0);
      // dP^T[key][m] = V dO^T: A=V (k=dv), B=dO rows (k=dv)
      f32x4_t dpt = {0.f, 0.f, 0.f, 0.f};
      dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          vf0, frag(sdO, ms * 16 + l16, kgrp * 8), dpt, 0, 0, 0);
      dpt = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          vf1, frag(sdO, ms * 16 + l16, kgrp * 8 + 32), dpt, 0, 0, 0);

      const int mcol = ms * 16 + l16;
      const float lse_m = sLse[mcol];
      const float del_m = sDelta[mcol];
      const int qrow = q0 + mcol;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int key = k0 + wkey0 + kgrp * 4 + r;
        const bool masked = (key >= Lk) || (causal && key > qrow);
        const float pt = masked ? 0.f : __expf(st[r] * scale - lse_m);
        sPT[(wkey0 + kgrp * 4 + r) * PITCH + mcol] = f2bf_rne(pt);
        dst_all[ms][r] = pt * (dpt[r] - del_m) * scale;
      }
    }

    // dV[key][dv] += P^T dO: A = P^T (k=m, sPT rows), B = dO^T rows (k=m)
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      bf16x8 ptf = frag(sPT, wkey0 + l16, kgrp * 8 + kc * 32);
#pragma unroll
      for (int ds_ = 0; ds_ < 4; ++ds_)
        acc_dv[ds_] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            ptf, frag(sdOT, ds_ * 16 + l16, kgrp * 8 + kc * 32),
            acc_dv[ds_], 0, 0, 0);
    }

    // overwrite sPT with dS^T (wave-local rows), then
    // dK[key][dim] += dS^T Q: A = dS^T (k=m), B = Q^T rows (k=m)
#pragma unroll
    for (int ms = 0; ms < 4; ++ms)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        sPT[(wkey0 + kgrp * 4 + r) * PITCH + ms * 16 + l16] =
            f2bf_rne(dst_all[ms][r]);

#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      bf16x8 dsf = frag(sPT, wkey0 + l16, kgrp * 8 + kc * 32);
#pragma unroll
      for (int ds_ = 0; ds_ < 4; ++ds_)
        acc_dk[ds_] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsf, frag(sQT, ds_ * 16 + l16, kgrp * 8 + kc * 32),
            acc_dk[ds_], 0, 0, 0);
    }
  }

  // store dK/dV rows (key-guarded)
  bf16_t* dkp = dK + (((long)b_ * Lk + k0) * H + h) * DHEAD;
  bf16_t* dvp = dV + (((long)b_ * Lk + k0) * H + h) * DHEAD;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int key = wkey0 + kgrp * 4 + r;
    if (k0 + key >= Lk) continue;
#pragma unroll
    for (int ds_ = 0; ds_ < 4; ++ds_) {
      dkp[(long)key * rs + ds_ * 16 + l16] = __float2bfloat16(acc_dk[ds_][r]);
      dvp[(long)key * rs + ds_ * 16 + l16] = __float2bfloat16(acc_dv[ds_][r]);
    }
  }
}

// ==========================================================================
// Backward dQ. grid (ceil(Lq/64), BH); wave owns 16 q rows.
// ==========================================================================
__global__ __launch_bounds__(256)
void attn_bwd_dq_kernel(const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
                        const bf16_t* __restrict__ v, const bf16_t* __restrict__ dO,
                        const float* __restrict__ lse, const float* __restrict__ delta,
                        bf16_t* __restrict__ dQ, int Lq, int Lk, int H,
                        float scale, int causal) {
  __shared__ short sQ[TILE * PITCH];
  __shared__ short sdO[TILE * PITCH];
  __shared__ short sK[TILE * PITCH];     // [key][dim]
  __shared__ short sKT[TILE * PITCH];    // [dim][key]
  __shared__ short sV[TILE * PITCH];     // [key][dim]
  __shared__ short sDS[TILE * PITCH];    // [qrow][key]

  const int bh = blockIdx.y;
  const int b_ = bh / H, h = bh % H;
  const int q0 = blockIdx.x * TILE;
  const long rs = (long)H * DHEAD;
  const bf16_t* qp = q + (((long)b_ * Lq + q0) * H + h) * DHEAD;
  const bf16_t* kp = k + ((long)b_ * Lk * H + h) * DHEAD;
  const bf16_t* vp = v + ((long)b_ * Lk * H + h) * DHEAD;
  const bf16_t* dop = dO + (((long)b_ * Lq + q0) * H + h) * DHEAD;

  load_tile(qp, rs, Lq - q0, sQ);
  load_tile(dop, rs, Lq - q0, sdO);

  const int lane = threadIdx.x & 63;
  const int wid = threadIdx.x >> 6;
  const int l16 = lane & 15;
  const int kgrp = lane >> 4;
  const int wrow0 = wid * 16;

  // per-lane row constants (rows m = kgrp*4 + r)
  float lse_r[4], del_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + wrow0 + kgrp * 4 + r;
    lse_r[r] = (qrow < Lq) ? lse[(long)bh * Lq + qrow] : 1e30f;
    del_r[r] = (qrow < Lq) ? delta[(long)bh * Lq + qrow] : 0.f;
  }

  f32x4_t acc_dq[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) acc_dq[i] = {0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? min(Lk, q0 + TILE) : Lk;

  for (int kv0 = 0; kv0 < kv_end; kv0 += TILE) {
    __syncthreads();
    load_tile(kp + (long)kv0 * rs, rs, Lk - kv0, sK);
    load_tile_T(kp + (long)kv0 * rs, rs, Lk - kv0, sKT);
    load_tile(vp + (long)kv0 * rs, rs, Lk - kv0, sV);
    __syncthreads();

    bf16x8 qf0 = frag(sQ, wrow0 + l16, kgrp * 8);
    bf16x8 qf1 = frag(sQ, wrow0 + l16, kgrp * 8 + 32);
    bf16x8 df0 = frag(sdO, wrow0 + l16, kgrp * 8);
    bf16x8 df1 = frag(sdO, wrow0 + l16, kgrp * 8 + 32);

#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
      // S[m][key] = Q K^T: A=Q (k=dim), B=K rows (k=dim)
      f32x4_t s = {0.f, 0.f, 0.f, 0.f};
      s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          qf0, frag(sK, ns * 16 + l16, kgrp * 8), s, 0, 0, 0);
      s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          qf1, frag(sK, ns * 16 + l16, kgrp * 8 + 32), s, 0, 0, 0);
      // dP[m][key] = dO V^T: A=dO (k=dv), B=V rows (k=dv)
      f32x4_t dp = {0.f, 0.f, 0.f, 0.f};
      dp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          df0, frag(sV, ns * 16 + l16, kgrp * 8), dp, 0, 0, 0);
      dp = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          df1, frag(sV, ns * 16 + l16, kgrp * 8 + 32), dp, 0, 0, 0);

      const int key = kv0 + ns * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int qrow = q0 + wrow0 + kgrp * 4 + r;
        const bool masked = (key >= Lk) || (causal && key > qrow);
        const float p = masked ? 0.f : __expf(s[r] * scale - lse_r[r]);
        sDS[(wrow0 + kgrp * 4 + r) * PITCH + ns * 16 + l16] =
            f2bf_rne(p * (dp[r] - del_r[r]) * scale);
      }
    }

    // dQ[m][dim] += dS K: A = dS (k=key, sDS rows), B = K^T rows (k=key)
#pragma unroll
    for (int kc = 0; kc < 2; ++kc) {
      bf16x8 dsf = frag(sDS, wrow0 + l16, kgrp * 8 + kc * 32);
#pragma unroll
      for (int ds_ = 0; ds_ < 4; ++ds_)
        acc_dq[ds_] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            dsf, frag(sKT, ds_ * 16 + l16, kgrp * 8 + kc * 32),
            acc_dq[ds_], 0, 0, 0);
    }
  }

  bf16_t* dqp = dQ + (((long)b_ * Lq + q0) * H + h) * DHEAD;
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int row = wrow0 + kgrp * 4 + r;
    if (q0 + row >= Lq) continue;
#pragma unroll
    for (int ds_ = 0; ds_ < 4; ++ds_)
      dqp[(long)row * rs + ds_ * 16 + l16] = __float2bfloat16(acc_dq[ds_][r]);
  }
}

// ==========================================================================
// mfma layout probe: C[16][16] = A[16][32] @ B[32][16], my frag conventions.
// Used by tests to pin the hardware fragment layout.
// ==========================================================================
__global__ void mfma_probe_kernel(const bf16_t* __restrict__ A,
                                  const bf16_t* __restrict__ B,
                                  float* __restrict__ C) {
  const int l = threadIdx.x & 63;
  bf16x8 a, b;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    a[j] = *reinterpret_cast<const short*>(A + (l & 15) * 32 + (l >> 4) * 8 + j);
    b[j] = *reinterpret_cast<const short*>(B + ((l >> 4) * 8 + j) * 16 + (l & 15));
  }
  f32x4_t c = {0.f, 0.f, 0.f, 0.f};
  c = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, c, 0, 0, 0);
#pragma unroll
  for (int r = 0; r < 4; ++r) C[((l >> 4) * 4 + r) * 16 + (l & 15)] = c[r];
}

// ==========================================================================
// Forward v3 (round-2 draft, DCR_ATTN_V3): same math as v1 with the
// guide's T14 async-STAGE split + T5 setprio applied to the L >= 1024
// regime where v1 trails AOTriton SDPA (182 vs 385 TF @1024):
//   * K/V tiles double-buffered; the NEXT tile's global loads are issued
//     into registers BEFORE this tile's MFMA/softmax phase (HBM latency
//     hides under compute), written to the alternate LDS buffer after it
//   * ONE barrier per K/V tile instead of two
//   * s_setprio(1) around the MFMA clusters (T5: phase-split scheduling)
//
// MEASURED (r02c5, bit-equal vs v1): 0.75-0.82x of v1 at L >= 256 —
// the second K/V buffer pushes LDS to 110 KB -> 1 block/CU (v1: 73 KB,
// 2 blocks/CU) and the lost wave-level overlap outweighs the staging
// pipeline, exactly like the conv v3 draft. NOT dispatched; kept as the
// documented negative result. The path to SDPA-class throughput at
// L >= 1024 is the restructured schedule (swapped QK^T, in-register
// softmax), not staging grafts on this structure.
// ==========================================================================
__global__ __launch_bounds__(256)
void attn_fwd_v3_kernel(const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
                        const bf16_t* __restrict__ v, bf16_t* __restrict__ o,
                        float* __restrict__ lse, int Lq, int Lk, int H,
                        float scale, int causal) {
  __shared__ short sQ[TILE * PITCH];
  __shared__ short sK[2][TILE * PITCH];
  __shared__ short sVT[2][TILE * PITCH];
  __shared__ short sP[TILE * PITCH];

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int q0 = blockIdx.x * TILE;
  const long qrs = (long)H * DHEAD;
  const bf16_t* qp = q + (((long)b * Lq + q0) * H + h) * DHEAD;
  const bf16_t* kp = k + ((long)b * Lk * H + h) * DHEAD;
  const bf16_t* vp = v + ((long)b * Lk * H + h) * DHEAD;
  bf16_t* op = o + ((long)b * Lq * H + h) * DHEAD;

  const int t = threadIdx.x;
  const int st_row = t >> 2;
  const int st_col = (t & 3) * 16;

  // register staging: K rows as 2 uint4, V rows as 16 shorts (pre-split
  // for the transposed write)
  uint4 ka, kb;
  short vv[16];

  auto issue_tile = [&](int kv0) {
    const int valid = Lk - kv0;
    ka = make_uint4(0, 0, 0, 0);
    kb = ka;
    if (st_row < valid) {
      const uint4* pk = reinterpret_cast<const uint4*>(
          kp + ((long)(kv0 + st_row)) * qrs + st_col);
      ka = pk[0];
      kb = pk[1];
      const uint4* pv = reinterpret_cast<const uint4*>(
          vp + ((long)(kv0 + st_row)) * qrs + st_col);
      *reinterpret_cast<uint4*>(vv) = pv[0];
      *reinterpret_cast<uint4*>(vv + 8) = pv[1];
    } else {
#pragma unroll
      for (int j = 0; j < 16; ++j) vv[j] = 0;
    }
  };
  auto write_tile = [&](int buf) {
    uint4* d = reinterpret_cast<uint4*>(sK[buf] + st_row * PITCH + st_col);
    d[0] = ka;
    d[1] = kb;
#pragma unroll
    for (int j = 0; j < 16; ++j)
      sVT[buf][(st_col + j) * PITCH + st_row] = vv[j];
  };

  load_tile(qp, qrs, Lq - q0, sQ);

  const int lane = t & 63;
  const int wid = t >> 6;
  const int l16 = lane & 15;
  const int kgrp = lane >> 4;
  const int wrow0 = wid * 16;

  float row_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  float row_sum[4] = {0.f, 0.f, 0.f, 0.f};
  f32x4_t acc_o[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) acc_o[i] = {0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? min(Lk, q0 + TILE) : Lk;

  issue_tile(0);
  write_tile(0);
  __syncthreads();

  int cur = 0;
  for (int kv0 = 0; kv0 < kv_end; kv0 += TILE) {
    const bool more = (kv0 + TILE < kv_end);
    if (more) issue_tile(kv0 + TILE);

    bf16x8 qf0 = frag(sQ, wrow0 + l16, kgrp * 8);
    bf16x8 qf1 = frag(sQ, wrow0 + l16, kgrp * 8 + 32);
    f32x4_t s_frag[4];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
      f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          qf0, frag(sK[cur], ns * 16 + l16, kgrp * 8), acc, 0, 0, 0);
      acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          qf1, frag(sK[cur], ns * 16 + l16, kgrp * 8 + 32), acc, 0, 0, 0);
      s_frag[ns] = acc;
    }
    __builtin_amdgcn_s_setprio(0);

    float tile_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
      const int key = kv0 + ns * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float sv = s_frag[ns][r] * scale;
        const int qrow = q0 + wrow0 + kgrp * 4 + r;
        if (key >= Lk || (causal && key > qrow)) sv = -1e30f;
        s_frag[ns][r] = sv;
        tile_max[r] = fmaxf(tile_max[r], sv);
      }
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float tm = qmax(tile_max[r]);
      float mnew = fmaxf(row_max[r], tm);
      alpha[r] = (mnew <= -1e29f) ? 1.f : __expf(row_max[r] - mnew);
      row_max[r] = mnew;
    }

    float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float pv = (s_frag[ns][r] <= -1e29f)
                       ? 0.f : __expf(s_frag[ns][r] - row_max[r]);
        psum[r] += pv;
        sP[(wrow0 + kgrp * 4 + r) * PITCH + ns * 16 + l16] = f2bf_rne(pv);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) row_sum[r] = row_sum[r] * alpha[r] + qsum(psum[r]);

#pragma unroll
    for (int ds_ = 0; ds_ < 4; ++ds_)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[ds_][r] *= alpha[r];

    bf16x8 pf0 = frag(sP, wrow0 + l16, kgrp * 8);
    bf16x8 pf1 = frag(sP, wrow0 + l16, kgrp * 8 + 32);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ds_ = 0; ds_ < 4; ++ds_) {
      acc_o[ds_] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          pf0, frag(sVT[cur], ds_ * 16 + l16, kgrp * 8), acc_o[ds_], 0, 0, 0);
      acc_o[ds_] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          pf1, frag(sVT[cur], ds_ * 16 + l16, kgrp * 8 + 32), acc_o[ds_], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);

    if (more) {
      // previous readers of buf^1 finished before the LAST barrier, so
      // the write needs no pre-barrier; the one barrier publishes it
      write_tile(cur ^ 1);
      __syncthreads();
      cur ^= 1;
    }
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + wrow0 + kgrp * 4 + r;
    if (qrow >= Lq) continue;
    const float inv = (row_sum[r] > 0.f) ? 1.f / row_sum[r] : 0.f;
#pragma unroll
    for (int ds_ = 0; ds_ < 4; ++ds_)
      op[(long)qrow * qrs + ds_ * 16 + l16] = __float2bfloat16(acc_o[ds_][r] * inv);
    if (l16 == 0 && lse != nullptr)
      lse[(long)bh * Lq + qrow] = row_max[r] + __logf(fmaxf(row_sum[r], 1e-30f));
  }
}

// ==========================================================================
// Forward v4 (round-2, DCR_ATTN_V4): restructured schedule for the
// L >= 256 self-attention shapes where the v1 structure trails AOTriton
// SDPA. Following the guide's 8-warp ladder:
//   * 8 waves x 32 q-rows = 256 q-rows per block; K/V staging amortizes
//     over 4x the rows of v1 and Q lives in registers.
//   * swapped QK^T — mfma_f32_32x32x16_bf16(A=K, B=Q) gives S^T[key][q]
//     with each lane holding ONE q column (l&31): the entire online
//     softmax is lane-local (31 fmax/adds + one lane^32 exchange), no
//     quarter-wave shuffle reductions and no sP LDS round-trip.
//   * P stays in registers: bf16 pack + v_permlane32_swap assembles the
//     PV B-fragments directly (guide T12 layout algebra).
//   * PV is also swapped — mfma(A=V^T, B=P) accumulates O^T[d][q] so the
//     alpha rescale stays lane-local too.
// Fragment maps (guide §3, gfx950): 32x32x16 A/B: lane l holds row l&31,
// k = (l>>5)*8 + j; C/D: col = l&31, row = (r&3) + 8*(r>>2) + 4*(l>>5).
// ==========================================================================
typedef __attribute__((ext_vector_type(16))) float f32x16_t;

__global__ __launch_bounds__(512)
void attn_fwd_v4_kernel(const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
                        const bf16_t* __restrict__ v, bf16_t* __restrict__ o,
                        float* __restrict__ lse, int Lq, int Lk, int H,
                        float scale, int causal) {
  __shared__ short sK[TILE * PITCH];    // [64 keys][72] (d contiguous)
  __shared__ short sVT[TILE * PITCH];   // [64 d][72]   (keys contiguous)

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int q0 = blockIdx.x * 256;
  const long qrs = (long)H * DHEAD;
  const bf16_t* kp = k + ((long)b * Lk * H + h) * DHEAD;
  const bf16_t* vp = v + ((long)b * Lk * H + h) * DHEAD;
  bf16_t* op = o + ((long)b * Lq * H + h) * DHEAD;

  const int t = threadIdx.x;
  const int wid = t >> 6;
  const int lane = t & 63;
  const int l32 = lane & 31;
  const int hi = lane >> 5;
  const int qrow = q0 + wid * 32 + l32;  // this lane's q row (all regs)

  // Q fragments in registers: qf[c] = Q[qrow][c*16 + hi*8 .. +8]
  bf16x8 qf[4];
  if (qrow < Lq) {
    const bf16_t* qp = q + (((long)b * Lq + qrow) * H + h) * DHEAD;
#pragma unroll
    for (int c = 0; c < 4; ++c)
      qf[c] = *reinterpret_cast<const bf16x8*>(qp + c * 16 + hi * 8);
  } else {
#pragma unroll
    for (int c = 0; c < 4; ++c) qf[c] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
  }

  // cooperative staging: 512 threads, 8 threads/row, one uint4 each
  const int st_row = t >> 3;
  const int st_col = (t & 7) * 8;

  float m_run = -1e30f, l_run = 0.f;
  f32x16_t oacc[2];
#pragma unroll
  for (int dt = 0; dt < 2; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) oacc[dt][r] = 0.f;

  const int kv_end = causal ? min(Lk, q0 + 256) : Lk;

  for (int kv0 = 0; kv0 < kv_end; kv0 += TILE) {
    __syncthreads();
    {
      const int valid = Lk - kv0;
      uint4 kv4 = make_uint4(0, 0, 0, 0);
      short vv[8];
#pragma unroll
      for (int j = 0; j < 8; ++j) vv[j] = 0;
      if (st_row < valid) {
        kv4 = *reinterpret_cast<const uint4*>(
            kp + (long)(kv0 + st_row) * qrs + st_col);
        *reinterpret_cast<uint4*>(vv) = *reinterpret_cast<const uint4*>(
            vp + (long)(kv0 + st_row) * qrs + st_col);
      }
      *reinterpret_cast<uint4*>(sK + st_row * PITCH + st_col) = kv4;
#pragma unroll
      for (int j = 0; j < 8; ++j)
        sVT[(st_col + j) * PITCH + st_row] = vv[j];
    }
    __syncthreads();

    // S^T = scale * (K Q^T): two 32-key subtiles, contraction D in 16s
    f32x16_t sacc[2];
#pragma unroll
    for (int st = 0; st < 2; ++st) {
      f32x16_t acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = 0.f;
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        bf16x8 kf = *reinterpret_cast<const bf16x8*>(
            sK + (st * 32 + l32) * PITCH + c * 16 + hi * 8);
        acc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[c], acc, 0, 0, 0);
      }
      sacc[st] = acc;
    }

    // mask + lane-local online softmax (this lane owns row qrow)
    float tmax = -1e30f;
#pragma unroll
    for (int st = 0; st < 2; ++st)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int key = kv0 + st * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        float sv = sacc[st][r] * scale;
        if (key >= Lk || (causal && key > qrow)) sv = -1e30f;
        sacc[st][r] = sv;
        tmax = fmaxf(tmax, sv);
      }
    tmax = fmaxf(tmax, __shfl_xor(tmax, 32, 64));
    const float mnew = fmaxf(m_run, tmax);
    const float alpha = (mnew <= -1e29f) ? 1.f : __expf(m_run - mnew);
    m_run = mnew;

    float psum = 0.f;
#pragma unroll
    for (int st = 0; st < 2; ++st)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float pv = (sacc[st][r] <= -1e29f) ? 0.f
                                           : __expf(sacc[st][r] - m_run);
        sacc[st][r] = pv;
        psum += pv;
      }
    psum += __shfl_xor(psum, 32, 64);
    l_run = l_run * alpha + psum;

#pragma unroll
    for (int dt = 0; dt < 2; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r) oacc[dt][r] *= alpha;

    // P -> bf16 fragments via pack + permlane32_swap, then O^T += V^T P
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      const int st = c >> 1;
      const int r8 = 8 * (c & 1);
      unsigned u[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) {
        __hip_bfloat16 lo = __float2bfloat16(sacc[st][r8 + 2 * i]);
        __hip_bfloat16 hi_ = __float2bfloat16(sacc[st][r8 + 2 * i + 1]);
        u[i] = (unsigned)*reinterpret_cast<unsigned short*>(&lo) |
               ((unsigned)*reinterpret_cast<unsigned short*>(&hi_) << 16);
      }
      auto s0 = __builtin_amdgcn_permlane32_swap(u[0], u[2], false, false);
      auto s1 = __builtin_amdgcn_permlane32_swap(u[1], u[3], false, false);
      unsigned w[4] = {(unsigned)s0[0], (unsigned)s1[0],
                       (unsigned)s0[1], (unsigned)s1[1]};
      bf16x8 pfrag = *reinterpret_cast<const bf16x8*>(w);
#pragma unroll
      for (int dt = 0; dt < 2; ++dt) {
        bf16x8 vf = *reinterpret_cast<const bf16x8*>(
            sVT + (dt * 32 + l32) * PITCH + c * 16 + hi * 8);
        oacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, pfrag,
                                                           oacc[dt], 0, 0, 0);
      }
    }
  }

  if (qrow >= Lq) return;
  const float inv = (l_run > 0.f) ? 1.f / l_run : 0.f;
#pragma unroll
  for (int dt = 0; dt < 2; ++dt)
#pragma unroll
    for (int rq = 0; rq < 4; ++rq) {
      const int d0 = dt * 32 + 8 * rq + 4 * hi;
      ushort4 pk;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        __hip_bfloat16 hv = __float2bfloat16(oacc[dt][rq * 4 + e] * inv);
        (&pk.x)[e] = *reinterpret_cast<unsigned short*>(&hv);
      }
      *reinterpret_cast<uint2*>(op + (long)qrow * qrs + d0) =
          *reinterpret_cast<uint2*>(&pk);
    }
  if (hi == 0 && lse != nullptr)
    lse[(long)bh * Lq + qrow] = m_run + __logf(fmaxf(l_run, 1e-30f));
}

// ==========================================================================
// Backward v4 (round-2 draft, DCR_ATTN_BWD_V4): the fwd-v4 schedule
// applied to the FlashAttention-2 backward. Backward needs no online
// softmax (P recomputes from the stored LSE), so both kernels keep the
// whole P / dS state in registers with the swapped-operand layout and
// the same pack+permlane fragment algebra fwd v4 verified:
//   * dq_v4:   8 waves x 32 q-rows; lane owns ONE q (lse/delta are
//     lane-local scalars); S^T/dP^T/dS^T all lane-local; dQ^T
//     accumulates like fwd's O^T. No LDS round-trip for P or dS.
//   * dkdv_v4: 8 waves x 32 keys; lane owns ONE key; per-q lse/delta
//     come from an LDS stage; dV/dK accumulate over q tiles.
// ==========================================================================

// pack a lane-local 64-wide fp32 axis (two 32-subtile reg files, crow
// layout) into the bf16 A/B fragment for contraction chunk c (16 wide):
// words = (s0[0], s1[0], s0[1], s1[1]) — fwd-v4-verified algebra.
__device__ __forceinline__ bf16x8 pack_frag64(const f32x16_t& t0,
                                              const f32x16_t& t1, int c) {
  const f32x16_t& t = (c >> 1) ? t1 : t0;
  const int r8 = 8 * (c & 1);
  unsigned u[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    __hip_bfloat16 lo = __float2bfloat16(t[r8 + 2 * i]);
    __hip_bfloat16 hi_ = __float2bfloat16(t[r8 + 2 * i + 1]);
    u[i] = (unsigned)*reinterpret_cast<unsigned short*>(&lo) |
           ((unsigned)*reinterpret_cast<unsigned short*>(&hi_) << 16);
  }
  auto s0 = __builtin_amdgcn_permlane32_swap(u[0], u[2], false, false);
  auto s1 = __builtin_amdgcn_permlane32_swap(u[1], u[3], false, false);
  unsigned w[4] = {(unsigned)s0[0], (unsigned)s1[0],
                   (unsigned)s0[1], (unsigned)s1[1]};
  return *reinterpret_cast<const bf16x8*>(w);
}

__global__ __launch_bounds__(512)
void attn_bwd_dq_v4_kernel(const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
                           const bf16_t* __restrict__ v, const bf16_t* __restrict__ dO,
                           const float* __restrict__ lse,
                           const float* __restrict__ delta,
                           bf16_t* __restrict__ dQ, int Lq, int Lk, int H,
                           float scale, int causal) {
  __shared__ short sK[TILE * PITCH];     // [key][dim]
  __shared__ short sKT[TILE * PITCH];    // [dim][key]
  __shared__ short sV[TILE * PITCH];     // [key][dim]

  const int bh = blockIdx.y;
  const int b_ = bh / H, h = bh % H;
  const int q0 = blockIdx.x * 256;
  const long rs = (long)H * DHEAD;
  const bf16_t* kp = k + ((long)b_ * Lk * H + h) * DHEAD;
  const bf16_t* vp = v + ((long)b_ * Lk * H + h) * DHEAD;

  const int t = threadIdx.x;
  const int wid = t >> 6;
  const int lane = t & 63;
  const int l32 = lane & 31;
  const int hi = lane >> 5;
  const int qrow = q0 + wid * 32 + l32;
  const bool valid = qrow < Lq;

  bf16x8 qf[4], of[4];
  float lse_q = 1e30f, del_q = 0.f;
  if (valid) {
    const bf16_t* qp = q + (((long)b_ * Lq + qrow) * H + h) * DHEAD;
    const bf16_t* dop = dO + (((long)b_ * Lq + qrow) * H + h) * DHEAD;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      qf[c] = *reinterpret_cast<const bf16x8*>(qp + c * 16 + hi * 8);
      of[c] = *reinterpret_cast<const bf16x8*>(dop + c * 16 + hi * 8);
    }
    lse_q = lse[(long)bh * Lq + qrow];
    del_q = delta[(long)bh * Lq + qrow];
  } else {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      qf[c] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      of[c] = qf[c];
    }
  }

  // staging: 512 threads, 8/row, one uint4 + 8-scalar transposed write
  const int st_row = t >> 3;
  const int st_col = (t & 7) * 8;

  f32x16_t dqacc[2];
#pragma unroll
  for (int dt = 0; dt < 2; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) dqacc[dt][r] = 0.f;

  const int kv_end = causal ? min(Lk, q0 + 256) : Lk;

  for (int kv0 = 0; kv0 < kv_end; kv0 += TILE) {
    __syncthreads();
    {
      const int vld = Lk - kv0;
      uint4 kv4 = make_uint4(0, 0, 0, 0), vv4 = kv4;
      if (st_row < vld) {
        kv4 = *reinterpret_cast<const uint4*>(
            kp + (long)(kv0 + st_row) * rs + st_col);
        vv4 = *reinterpret_cast<const uint4*>(
            vp + (long)(kv0 + st_row) * rs + st_col);
      }
      *reinterpret_cast<uint4*>(sK + st_row * PITCH + st_col) = kv4;
      *reinterpret_cast<uint4*>(sV + st_row * PITCH + st_col) = vv4;
      const short* kk_ = reinterpret_cast<const short*>(&kv4);
#pragma unroll
      for (int j = 0; j < 8; ++j)
        sKT[(st_col + j) * PITCH + st_row] = kk_[j];
    }
    __syncthreads();

    f32x16_t pt[2], dpt[2];
#pragma unroll
    for (int st = 0; st < 2; ++st) {
      f32x16_t sacc, dacc;
#pragma unroll
      for (int r = 0; r < 16; ++r) { sacc[r] = 0.f; dacc[r] = 0.f; }
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        bf16x8 kf = *reinterpret_cast<const bf16x8*>(
            sK + (st * 32 + l32) * PITCH + c * 16 + hi * 8);
        bf16x8 vf = *reinterpret_cast<const bf16x8*>(
            sV + (st * 32 + l32) * PITCH + c * 16 + hi * 8);
        sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf[c], sacc, 0, 0, 0);
        dacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, of[c], dacc, 0, 0, 0);
      }
      pt[st] = sacc;
      dpt[st] = dacc;
    }

    // P^T from lse; dS^T = P^T (dP^T - delta) * scale — all lane-local
#pragma unroll
    for (int st = 0; st < 2; ++st)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int key = kv0 + st * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const bool masked = !valid || key >= Lk || (causal && key > qrow);
        const float p = masked ? 0.f : __expf(pt[st][r] * scale - lse_q);
        pt[st][r] = p * (dpt[st][r] - del_q) * scale;   // now dS^T
      }

    // dQ^T[d][q] += K^T dS^T
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      bf16x8 dsf = pack_frag64(pt[0], pt[1], c);
#pragma unroll
      for (int dt = 0; dt < 2; ++dt) {
        bf16x8 ktf = *reinterpret_cast<const bf16x8*>(
            sKT + (dt * 32 + l32) * PITCH + c * 16 + hi * 8);
        dqacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ktf, dsf,
                                                            dqacc[dt], 0, 0, 0);
      }
    }
  }

  if (!valid) return;
  bf16_t* dqp = dQ + (((long)b_ * Lq + qrow) * H + h) * DHEAD;
#pragma unroll
  for (int dt = 0; dt < 2; ++dt)
#pragma unroll
    for (int rq = 0; rq < 4; ++rq) {
      const int d0 = dt * 32 + 8 * rq + 4 * hi;
      ushort4 pk;
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        __hip_bfloat16 hv = __float2bfloat16(dqacc[dt][rq * 4 + e]);
        (&pk.x)[e] = *reinterpret_cast<unsigned short*>(&hv);
      }
      *reinterpret_cast<uint2*>(dqp + d0) = *reinterpret_cast<uint2*>(&pk);
    }
}

__global__ __launch_bounds__(512)
void attn_bwd_dkdv_v4_kernel(const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
                             const bf16_t* __restrict__ v, const bf16_t* __restrict__ dO,
                             const float* __restrict__ lse,
                             const float* __restrict__ delta,
                             bf16_t* __restrict__ dK, bf16_t* __restrict__ dV,
                             int Lq, int Lk, int H, float scale, int causal) {
  __shared__ short sQ[TILE * PITCH];     // [qrow][dim]
  __shared__ short sQT[TILE * PITCH];    // [dim][qrow]
  __shared__ short sdO[TILE * PITCH];    // [qrow][dim]
  __shared__ short sdOT[TILE * PITCH];   // [dim][qrow]
  __shared__ float sLse[TILE];
  __shared__ float sDelta[TILE];

  const int bh = blockIdx.y;
  const int b_ = bh / H, h = bh % H;
  const int k0 = blockIdx.x * 256;
  const long rs = (long)H * DHEAD;
  const bf16_t* qp = q + ((long)b_ * Lq * H + h) * DHEAD;
  const bf16_t* dop = dO + ((long)b_ * Lq * H + h) * DHEAD;
  const float* lsep = lse + (long)bh * Lq;
  const float* delp = delta + (long)bh * Lq;

  const int t = threadIdx.x;
  const int wid = t >> 6;
  const int lane = t & 63;
  const int l32 = lane & 31;
  const int hi = lane >> 5;
  const int key = k0 + wid * 32 + l32;
  const bool kvalid = key < Lk;

  bf16x8 kf[4], vf[4];
  if (kvalid) {
    const bf16_t* kp = k + (((long)b_ * Lk + key) * H + h) * DHEAD;
    const bf16_t* vp = v + (((long)b_ * Lk + key) * H + h) * DHEAD;
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      kf[c] = *reinterpret_cast<const bf16x8*>(kp + c * 16 + hi * 8);
      vf[c] = *reinterpret_cast<const bf16x8*>(vp + c * 16 + hi * 8);
    }
  } else {
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      kf[c] = bf16x8{0, 0, 0, 0, 0, 0, 0, 0};
      vf[c] = kf[c];
    }
  }

  const int st_row = t >> 3;
  const int st_col = (t & 7) * 8;

  f32x16_t dkacc[2], dvacc[2];
#pragma unroll
  for (int dt = 0; dt < 2; ++dt)
#pragma unroll
    for (int r = 0; r < 16; ++r) { dkacc[dt][r] = 0.f; dvacc[dt][r] = 0.f; }

  const int q_start = causal ? (k0 & ~(TILE - 1)) : 0;

  for (int q0 = q_start; q0 < Lq; q0 += TILE) {
    __syncthreads();
    {
      const int vld = Lq - q0;
      uint4 q4 = make_uint4(0, 0, 0, 0), o4 = q4;
      if (st_row < vld) {
        q4 = *reinterpret_cast<const uint4*>(
            qp + (long)(q0 + st_row) * rs + st_col);
        o4 = *reinterpret_cast<const uint4*>(
            dop + (long)(q0 + st_row) * rs + st_col);
      }
      *reinterpret_cast<uint4*>(sQ + st_row * PITCH + st_col) = q4;
      *reinterpret_cast<uint4*>(sdO + st_row * PITCH + st_col) = o4;
      const short* qq_ = reinterpret_cast<const short*>(&q4);
      const short* oo_ = reinterpret_cast<const short*>(&o4);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        sQT[(st_col + j) * PITCH + st_row] = qq_[j];
        sdOT[(st_col + j) * PITCH + st_row] = oo_[j];
      }
      if (t < TILE) {
        const int qr = q0 + t;
        sLse[t] = (qr < Lq) ? lsep[qr] : 1e30f;
        sDelta[t] = (qr < Lq) ? delp[qr] : 0.f;
      }
    }
    __syncthreads();

    // S'^T[key][q] and dP'^T[key][q] (lane = one key, q in regs)
    f32x16_t pt[2], dpt[2];
#pragma unroll
    for (int st = 0; st < 2; ++st) {
      f32x16_t sacc, dacc;
#pragma unroll
      for (int r = 0; r < 16; ++r) { sacc[r] = 0.f; dacc[r] = 0.f; }
#pragma unroll
      for (int c = 0; c < 4; ++c) {
        bf16x8 qfr = *reinterpret_cast<const bf16x8*>(
            sQ + (st * 32 + l32) * PITCH + c * 16 + hi * 8);
        bf16x8 ofr = *reinterpret_cast<const bf16x8*>(
            sdO + (st * 32 + l32) * PITCH + c * 16 + hi * 8);
        // A = Q rows (q), B = K lane frags -> C[q][key]: lane key-col?
        // No: A rows become C rows; we want C[key][q], so A = K? K is
        // in registers (kf) which are this lane's key only — use the
        // swapped form: mfma(A=Q_row_frag? ) — instead compute
        // C[q-subtile][key]^T by A=Q, B=kf: C[qrow][keycol]: lane holds
        // key-col l32?? kf holds THIS lane's key rows... B-frag rows
        // must be the output-col axis across lanes: kf[c] per lane IS
        // row l&31 of the B operand when B = K tile — but kf was loaded
        // for key = wid*32+l32, matching B rows = this wave's 32 keys.
        sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qfr, kf[c], sacc, 0, 0, 0);
        dacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(ofr, vf[c], dacc, 0, 0, 0);
      }
      // C[q][key]: lane holds key-col l32 (ours), q rows in regs
      pt[st] = sacc;
      dpt[st] = dacc;
    }

#pragma unroll
    for (int st = 0; st < 2; ++st)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int qr = q0 + st * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        const int qloc = qr - q0;
        const float lse_q = sLse[qloc];
        const float del_q = sDelta[qloc];
        const bool masked = !kvalid || qr >= Lq || (causal && key > qr);
        const float p = masked ? 0.f : __expf(pt[st][r] * scale - lse_q);
        pt[st][r] = p;                         // P'^T
        dpt[st][r] = p * (dpt[st][r] - del_q) * scale;  // dS'^T
      }

    // dV[key][d] += P'^T dO ; dK[key][d] += dS'^T Q   (contraction q)
#pragma unroll
    for (int c = 0; c < 4; ++c) {
      bf16x8 pf = pack_frag64(pt[0], pt[1], c);
      bf16x8 dsf = pack_frag64(dpt[0], dpt[1], c);
#pragma unroll
      for (int dt = 0; dt < 2; ++dt) {
        bf16x8 dotf = *reinterpret_cast<const bf16x8*>(
            sdOT + (dt * 32 + l32) * PITCH + c * 16 + hi * 8);
        bf16x8 qtf = *reinterpret_cast<const bf16x8*>(
            sQT + (dt * 32 + l32) * PITCH + c * 16 + hi * 8);
        dvacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pf, dotf,
                                                            dvacc[dt], 0, 0, 0);
        dkacc[dt] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dsf, qtf,
                                                            dkacc[dt], 0, 0, 0);
      }
    }
  }

  // C[key][d]: lane holds d-col?? dvacc came from mfma(A=P'^T[key][q],
  // B=dO^T[d][q]) -> C rows = key (A rows across lanes: l&31 = THIS
  // wave's key block rows), cols = d: lane holds col d = l32, rows
  // key = wkey + crow(r, hi). Store column-wise (scalar stores).
  if (true) {
    bf16_t* dkp = dK + ((long)b_ * Lk * H + h) * DHEAD;
    bf16_t* dvp = dV + ((long)b_ * Lk * H + h) * DHEAD;
#pragma unroll
    for (int dt = 0; dt < 2; ++dt)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        const int krow = k0 + wid * 32 + (r & 3) + 8 * (r >> 2) + 4 * hi;
        if (krow >= Lk) continue;
        const int d = dt * 32 + l32;
        dkp[(long)krow * rs + d] = __float2bfloat16(dkacc[dt][r]);
        dvp[(long)krow * rs + d] = __float2bfloat16(dvacc[dt][r]);
      }
  }
}

// ==========================================================================
// Generalized-head-dim forward (SD-1.4 / sd_mitigation parity: head_dim
// 40/80/160 — /root/reference/sd_mitigation.py:46; inference-only, so no
// backward). Same tile discipline as attn_fwd_kernel with the head dim
// padded to DP = ceil(D/32)*32 (zero-padded contraction contributes 0 to
// QK^T) and LDS pitch DP+8 bf16 — every (DP+8)*2 B row stride lands the
// 16 rows of a ds_read_b128 lane group on 16 distinct banks (stride/4
// mod 64 has gcd 4 with 64 for DP in {64,96,160}).
// ==========================================================================
template <int D, int DP, int GP>
__global__ __launch_bounds__(256)
void attn_fwd_gen_kernel(const bf16_t* __restrict__ q, const bf16_t* __restrict__ k,
                         const bf16_t* __restrict__ v, bf16_t* __restrict__ o,
                         float* __restrict__ lse, int Lq, int Lk, int H,
                         float scale, int causal) {
  __shared__ short sQ[TILE * GP];
  __shared__ short sK[TILE * GP];
  __shared__ short sVT[DP * PITCH];     // [dim][key], key pitch 72
  __shared__ short sP[TILE * PITCH];

  const int bh = blockIdx.y;
  const int b = bh / H, h = bh % H;
  const int q0 = blockIdx.x * TILE;
  const long rs = (long)H * D;            // [B,L,H,D] row stride
  const bf16_t* qp = q + (((long)b * Lq + q0) * H + h) * D;
  const bf16_t* kp = k + ((long)b * Lk * H + h) * D;
  const bf16_t* vp = v + ((long)b * Lk * H + h) * D;
  bf16_t* op = o + ((long)b * Lq * H + h) * D;

  const int t = threadIdx.x;
  // cooperative [64 rows][DP] load, 4 threads/row, 8-elem units (D%8==0)
  auto load_rows = [&](const bf16_t* src, int valid, short* dst) {
    const int row = t >> 2;
    const bf16_t* sp_ = src + (long)row * rs;
#pragma unroll
    for (int c = (t & 3) * 8; c < DP; c += 32) {
      uint4 a = make_uint4(0, 0, 0, 0);
      if (row < valid && c + 8 <= D)
        a = *reinterpret_cast<const uint4*>(sp_ + c);
      *reinterpret_cast<uint4*>(dst + row * GP + c) = a;
    }
  };
  auto load_vt = [&](const bf16_t* src, int valid) {
    const int row = t >> 2;
    const bf16_t* sp_ = src + (long)row * rs;
#pragma unroll
    for (int c0 = (t & 3) * 8; c0 < DP; c0 += 32) {
      short vv[8];
      if (row < valid && c0 + 8 <= D) {
        *reinterpret_cast<uint4*>(vv) = *reinterpret_cast<const uint4*>(sp_ + c0);
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) vv[j] = 0;
      }
#pragma unroll
      for (int j = 0; j < 8; ++j) sVT[(c0 + j) * PITCH + row] = vv[j];
    }
  };

  load_rows(qp, Lq - q0, sQ);

  const int lane = t & 63;
  const int wid = t >> 6;
  const int l16 = lane & 15;
  const int kgrp = lane >> 4;
  const int wrow0 = wid * 16;
  constexpr int NK = DP / 32;             // QK^T contraction chunks
  constexpr int ND = DP / 16;             // PV output d-subtiles

  float row_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  float row_sum[4] = {0.f, 0.f, 0.f, 0.f};
  f32x4_t acc_o[ND];
#pragma unroll
  for (int i = 0; i < ND; ++i) acc_o[i] = {0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? min(Lk, q0 + TILE) : Lk;

  for (int kv0 = 0; kv0 < kv_end; kv0 += TILE) {
    __syncthreads();
    load_rows(kp + (long)kv0 * rs, Lk - kv0, sK);
    load_vt(vp + (long)kv0 * rs, Lk - kv0);
    __syncthreads();

    bf16x8 qf[NK];
#pragma unroll
    for (int ck = 0; ck < NK; ++ck)
      qf[ck] = *reinterpret_cast<const bf16x8*>(
          sQ + (wrow0 + l16) * GP + ck * 32 + kgrp * 8);
    f32x4_t s_frag[4];
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
      f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
      for (int ck = 0; ck < NK; ++ck) {
        bf16x8 kf = *reinterpret_cast<const bf16x8*>(
            sK + (ns * 16 + l16) * GP + ck * 32 + kgrp * 8);
        acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qf[ck], kf, acc, 0, 0, 0);
      }
      s_frag[ns] = acc;
    }

    float tile_max[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
      const int key = kv0 + ns * 16 + l16;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float sv = s_frag[ns][r] * scale;
        const int qrow = q0 + wrow0 + kgrp * 4 + r;
        if (key >= Lk || (causal && key > qrow)) sv = -1e30f;
        s_frag[ns][r] = sv;
        tile_max[r] = fmaxf(tile_max[r], sv);
      }
    }
    float alpha[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      float tm = qmax(tile_max[r]);
      float mnew = fmaxf(row_max[r], tm);
      alpha[r] = (mnew <= -1e29f) ? 1.f : __expf(row_max[r] - mnew);
      row_max[r] = mnew;
    }

    float psum[4] = {0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ns = 0; ns < 4; ++ns) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        float pv = (s_frag[ns][r] <= -1e29f)
                       ? 0.f : __expf(s_frag[ns][r] - row_max[r]);
        psum[r] += pv;
        sP[(wrow0 + kgrp * 4 + r) * PITCH + ns * 16 + l16] = f2bf_rne(pv);
      }
    }
#pragma unroll
    for (int r = 0; r < 4; ++r) row_sum[r] = row_sum[r] * alpha[r] + qsum(psum[r]);

#pragma unroll
    for (int ds_ = 0; ds_ < ND; ++ds_)
#pragma unroll
      for (int r = 0; r < 4; ++r) acc_o[ds_][r] *= alpha[r];

    bf16x8 pf0 = frag(sP, wrow0 + l16, kgrp * 8);
    bf16x8 pf1 = frag(sP, wrow0 + l16, kgrp * 8 + 32);
#pragma unroll
    for (int ds_ = 0; ds_ < ND; ++ds_) {
      acc_o[ds_] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          pf0, frag(sVT, ds_ * 16 + l16, kgrp * 8), acc_o[ds_], 0, 0, 0);
      acc_o[ds_] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
          pf1, frag(sVT, ds_ * 16 + l16, kgrp * 8 + 32), acc_o[ds_], 0, 0, 0);
    }
  }

#pragma unroll
  for (int r = 0; r < 4; ++r) {
    const int qrow = q0 + wrow0 + kgrp * 4 + r;
    if (qrow >= Lq) continue;
    const float inv = (row_sum[r] > 0.f) ? 1.f / row_sum[r] : 0.f;
#pragma unroll
    for (int ds_ = 0; ds_ < ND; ++ds_) {
      const int d = ds_ * 16 + l16;
      if (d < D)
        op[(long)qrow * rs + d] = __float2bfloat16(acc_o[ds_][r] * inv);
    }
    if (l16 == 0 && lse != nullptr)
      lse[(long)bh * Lq + qrow] = row_max[r] + __logf(fmaxf(row_sum[r], 1e-30f));
  }
}

}  // namespace dcr_attn

// ==========================================================================
// Host launchers
// ==========================================================================
#include "dcr_launchers.h"

namespace dcr {

void attn_fwd_launch(const void* q, const void* k, const void* v, void* o,
                     float* lse, int BH, int Lq, int Lk, int H, float scale,
                     bool causal, hipStream_t s) {
  dim3 grid((Lq + TILE - 1) / TILE, BH), block(256);
  hipLaunchKernelGGL(dcr_attn::attn_fwd_kernel, grid, block, 0, s,
                     (const dcr_attn::bf16_t*)q, (const dcr_attn::bf16_t*)k,
                     (const dcr_attn::bf16_t*)v, (dcr_attn::bf16_t*)o, lse,
                     Lq, Lk, H, scale, causal ? 1 : 0);
}

// v2: masked-tail MFMA skip (round-2 draft; not dispatched)
void attn_fwd_v2_launch(const void* q, const void* k, const void* v, void* o,
                        float* lse, int BH, int Lq, int Lk, int H, float scale,
                        bool causal, hipStream_t s) {
  dim3 grid((Lq + TILE - 1) / TILE, BH), block(256);
  hipLaunchKernelGGL(dcr_attn::attn_fwd_v2_kernel, grid, block, 0, s,
                     (const dcr_attn::bf16_t*)q, (const dcr_attn::bf16_t*)k,
                     (const dcr_attn::bf16_t*)v, (dcr_attn::bf16_t*)o, lse,
                     Lq, Lk, H, scale, causal ? 1 : 0);
}

// v3: T14 async-stage + one barrier/tile + setprio (round-2 draft)
void attn_fwd_v3_launch(const void* q, const void* k, const void* v, void* o,
                        float* lse, int BH, int Lq, int Lk, int H, float scale,
                        bool causal, hipStream_t s) {
  dim3 grid((Lq + TILE - 1) / TILE, BH), block(256);
  hipLaunchKernelGGL(dcr_attn::attn_fwd_v3_kernel, grid, block, 0, s,
                     (const dcr_attn::bf16_t*)q, (const dcr_attn::bf16_t*)k,
                     (const dcr_attn::bf16_t*)v, (dcr_attn::bf16_t*)o, lse,
                     Lq, Lk, H, scale, causal ? 1 : 0);
}

// v4: swapped-QK^T in-register-softmax schedule (round-2)
void attn_fwd_v4_launch(const void* q, const void* k, const void* v, void* o,
                        float* lse, int BH, int Lq, int Lk, int H, float scale,
                        bool causal, hipStream_t s) {
  dim3 grid((Lq + 255) / 256, BH), block(512);
  hipLaunchKernelGGL(dcr_attn::attn_fwd_v4_kernel, grid, block, 0, s,
                     (const dcr_attn::bf16_t*)q, (const dcr_attn::bf16_t*)k,
                     (const dcr_attn::bf16_t*)v, (dcr_attn::bf16_t*)o, lse,
                     Lq, Lk, H, scale, causal ? 1 : 0);
}

void attn_bwd_launch(const void* q, const void* k, const void* v,
                     const void* o, const void* dO, const float* lse,
                     float* delta, void* dQ, void* dK, void* dV, int BH,
                     int Lq, int Lk, int H, float scale, bool causal,
                     hipStream_t s) {
  long rows = (long)BH * Lq;
  dim3 gd((rows + 255) / 256), bd(256);
  hipLaunchKernelGGL(dcr_attn::attn_bwd_delta_kernel, gd, bd, 0, s,
                     (const dcr_attn::bf16_t*)dO, (const dcr_attn::bf16_t*)o,
                     delta, rows, Lq, H);
  dim3 g1((Lk + TILE - 1) / TILE, BH), b1(256);
  hipLaunchKernelGGL(dcr_attn::attn_bwd_dkdv_kernel, g1, b1, 0, s,
                     (const dcr_attn::bf16_t*)q, (const dcr_attn::bf16_t*)k,
                     (const dcr_attn::bf16_t*)v, (const dcr_attn::bf16_t*)dO,
                     lse, delta, (dcr_attn::bf16_t*)dK, (dcr_attn::bf16_t*)dV,
                     Lq, Lk, H, scale, causal ? 1 : 0);
  dim3 g2((Lq + TILE - 1) / TILE, BH), b2(256);
  hipLaunchKernelGGL(dcr_attn::attn_bwd_dq_kernel, g2, b2, 0, s,
                     (const dcr_attn::bf16_t*)q, (const dcr_attn::bf16_t*)k,
                     (const dcr_attn::bf16_t*)v, (const dcr_attn::bf16_t*)dO,
                     lse, delta, (dcr_attn::bf16_t*)dQ, Lq, Lk, H, scale,
                     causal ? 1 : 0);
}

// generalized head-dim forward (SD-1.4 40/80/160; inference-only)
void attn_fwd_gen_launch(const void* q, const void* k, const void* v, void* o,
                         float* lse, int BH, int Lq, int Lk, int H, int D,
                         float scale, bool causal, hipStream_t s) {
  dim3 grid((Lq + TILE - 1) / TILE, BH), block(256);
  const auto* qq = (const dcr_attn::bf16_t*)q;
  const auto* kk = (const dcr_attn::bf16_t*)k;
  const auto* vv = (const dcr_attn::bf16_t*)v;
  auto* oo = (dcr_attn::bf16_t*)o;
  const int c = causal ? 1 : 0;
  switch (D) {
    case 40:
      hipLaunchKernelGGL((dcr_attn::attn_fwd_gen_kernel<40, 64, 72>), grid,
                         block, 0, s, qq, kk, vv, oo, lse, Lq, Lk, H, scale, c);
      break;
    case 80:
      hipLaunchKernelGGL((dcr_attn::attn_fwd_gen_kernel<80, 96, 104>), grid,
                         block, 0, s, qq, kk, vv, oo, lse, Lq, Lk, H, scale, c);
      break;
    case 160:
      hipLaunchKernelGGL((dcr_attn::attn_fwd_gen_kernel<160, 160, 168>), grid,
                         block, 0, s, qq, kk, vv, oo, lse, Lq, Lk, H, scale, c);
      break;
    default:
      break;  // binding guards D
  }
}

// backward v4 (swapped-operand schedule; DCR_ATTN_BWD_V4 draft)
void attn_bwd_v4_launch(const void* q, const void* k, const void* v,
                        const void* o, const void* dO, const float* lse,
                        float* delta, void* dQ, void* dK, void* dV, int BH,
                        int Lq, int Lk, int H, float scale, bool causal,
                        hipStream_t s) {
  long rows = (long)BH * Lq;
  dim3 gd((rows + 255) / 256), bd(256);
  hipLaunchKernelGGL(dcr_attn::attn_bwd_delta_kernel, gd, bd, 0, s,
                     (const dcr_attn::bf16_t*)dO, (const dcr_attn::bf16_t*)o,
                     delta, rows, Lq, H);
  dim3 g1((Lk + 255) / 256, BH), b1(512);
  hipLaunchKernelGGL(dcr_attn::attn_bwd_dkdv_v4_kernel, g1, b1, 0, s,
                     (const dcr_attn::bf16_t*)q, (const dcr_attn::bf16_t*)k,
                     (const dcr_attn::bf16_t*)v, (const dcr_attn::bf16_t*)dO,
                     lse, delta, (dcr_attn::bf16_t*)dK, (dcr_attn::bf16_t*)dV,
                     Lq, Lk, H, scale, causal ? 1 : 0);
  dim3 g2((Lq + 255) / 256, BH), b2(512);
  hipLaunchKernelGGL(dcr_attn::attn_bwd_dq_v4_kernel, g2, b2, 0, s,
                     (const dcr_attn::bf16_t*)q, (const dcr_attn::bf16_t*)k,
                     (const dcr_attn::bf16_t*)v, (const dcr_attn::bf16_t*)dO,
                     lse, delta, (dcr_attn::bf16_t*)dQ, Lq, Lk, H, scale,
                     causal ? 1 : 0);
}

void mfma_probe_launch(const void* A, const void* B, float* C, hipStream_t s) {
  hipLaunchKernelGGL(dcr_attn::mfma_probe_kernel, dim3(1), dim3(64), 0, s,
                     (const dcr_attn::bf16_t*)A, (const dcr_attn::bf16_t*)B, C);
}

}  // namespace dcr
