// GroupNorm(+fused SiLU) for channels_last (NHWC) tensors on MI355X.
//
// Rationale: MIOpen's fast igemm conv kernels are NHWC; with NCHW models
// every conv pays batched_transpose round-trips (profiles/prof4: ~5.6
// ms/step). channels_last removes them — but then GroupNorm must reduce
// over a strided [HW, Cg] slab. Design:
//   stats:   grid (row-chunks, N); threads own CHANNELS (consecutive c
//            => coalesced), register-accumulate over the chunk's rows,
//            LDS per-channel sums -> per-group partials -> global atomics
//            into a [N, G, 2] fp32 workspace. No per-element atomics.
//   finalize:[N*G] -> mean/rstd.
//   apply:   flat vecV elementwise (fully coalesced NHWC walk).
// Backward mirrors it (stats also emit per-channel dw/db).
//
// All four streaming kernels are templated on the vector width V: 8
// bf16/f16 elements = 16 B/lane (the HBM3E coalescing sweet spot, guide
// G13) whenever C % 8 == 0 (every SD-2.1/VAE channel count), else 4.

#include "dcr_common.h"

using namespace dcr;

namespace dcr_nhwc {

template <typename T, int V>
__device__ __forceinline__ void loadv(const T* __restrict__ p, float (&o)[V]) {
  if constexpr (V == 8) {
    f32x8 v = load8<T>(p);
#pragma unroll
    for (int k = 0; k < 4; ++k) { o[k] = (&v.lo.x)[k]; o[4 + k] = (&v.hi.x)[k]; }
  } else {
    f32x4 v = load4<T>(p);
#pragma unroll
    for (int k = 0; k < 4; ++k) o[k] = (&v.x)[k];
  }
}

template <typename T, int V>
__device__ __forceinline__ void storev(T* __restrict__ p, const float (&i)[V]) {
  if constexpr (V == 8) {
    f32x8 v;
#pragma unroll
    for (int k = 0; k < 4; ++k) { (&v.lo.x)[k] = i[k]; (&v.hi.x)[k] = i[4 + k]; }
    store8<T>(p, v);
  } else {
    f32x4 v;
#pragma unroll
    for (int k = 0; k < 4; ++k) (&v.x)[k] = i[k];
    store4<T>(p, v);
  }
}

// --------------------------------------------------------------- stats fwd
// ws is a per-chunk partial slab [chunks, N, G, 2] written with PLAIN
// stores (no zero-init, no atomics — deterministic); gn_finalize_kernel
// reduces over chunks.
template <typename T, int V>
__global__ void gn_nhwc_stats_kernel(const T* __restrict__ x, float* __restrict__ ws,
                                     int N, int R, int C, int G, int rows_per_blk) {
  extern __shared__ float smem[];          // [2*C]
  float* s1 = smem;
  float* s2 = smem + C;
  const int n = blockIdx.y;
  const int r0 = blockIdx.x * rows_per_blk;
  const int r1 = min(R, r0 + rows_per_blk);
  const long base = (long)n * R * C;
  const int Cg = C / G;

  if (C >= (int)blockDim.x * V) {
    // wide-C: thread t owns channels [V*t, V*t+V) per stripe (full util)
    for (int c0 = threadIdx.x * V; c0 < C; c0 += blockDim.x * V) {
      float a[V], b[V], v[V];
#pragma unroll
      for (int k = 0; k < V; ++k) { a[k] = 0.f; b[k] = 0.f; }
      for (int r = r0; r < r1; ++r) {
        loadv<T, V>(x + base + (long)r * C + c0, v);
#pragma unroll
        for (int k = 0; k < V; ++k) { a[k] += v[k]; b[k] += v[k] * v[k]; }
      }
#pragma unroll
      for (int k = 0; k < V; ++k) { s1[c0 + k] = a[k]; s2[c0 + k] = b[k]; }
    }
    __syncthreads();
  } else {
    // narrow-C: split threads over (channel-vec, row group) so all lanes
    // stay busy; combine row groups via LDS atomics.
    for (int c = threadIdx.x; c < 2 * C; c += blockDim.x) smem[c] = 0.f;
    __syncthreads();
    const int tpr = C / V;
    const int rpar = (int)blockDim.x / tpr;
    const int cidx = (threadIdx.x % tpr) * V;
    const int rgrp = threadIdx.x / tpr;
    if (rgrp < rpar) {
      float a[V], b[V], v[V];
#pragma unroll
      for (int k = 0; k < V; ++k) { a[k] = 0.f; b[k] = 0.f; }
      for (int r = r0 + rgrp; r < r1; r += rpar) {
        loadv<T, V>(x + base + (long)r * C + cidx, v);
#pragma unroll
        for (int k = 0; k < V; ++k) { a[k] += v[k]; b[k] += v[k] * v[k]; }
      }
#pragma unroll
      for (int k = 0; k < V; ++k) {
        atomicAdd(&s1[cidx + k], a[k]);
        atomicAdd(&s2[cidx + k], b[k]);
      }
    }
    __syncthreads();
  }
  float* slot = ws + (((long)blockIdx.x * N + n) * G) * 2;
  for (int g = threadIdx.x; g < G; g += blockDim.x) {
    float a = 0.f, b = 0.f;
    for (int c = g * Cg; c < (g + 1) * Cg; ++c) { a += s1[c]; b += s2[c]; }
    slot[(long)g * 2] = a;
    slot[(long)g * 2 + 1] = b;
  }
}

__global__ void gn_finalize_kernel(const float* __restrict__ ws,
                                   float* __restrict__ mean, float* __restrict__ rstd,
                                   long NG, int chunks, float inv_L, float eps) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= NG) return;
  float s1 = 0.f, s2 = 0.f;
  for (int cx = 0; cx < chunks; ++cx) {
    s1 += ws[((long)cx * NG + i) * 2];
    s2 += ws[((long)cx * NG + i) * 2 + 1];
  }
  float m = s1 * inv_L;
  float var = fmaxf(s2 * inv_L - m * m, 0.f);
  mean[i] = m;
  rstd[i] = rsqrtf(var + eps);
}

// reduce the bwd partial slabs: group sums [chunks, N, G, 2] -> [N*G, 2]
// and per-channel dw/db [chunks*N, 2C] -> dw[C], db[C]
__global__ void gn_bwd_finalize_groups_kernel(const float* __restrict__ ws,
                                              float* __restrict__ out, long NG,
                                              int chunks) {
  long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= NG) return;
  float s1 = 0.f, s2 = 0.f;
  for (int cx = 0; cx < chunks; ++cx) {
    s1 += ws[((long)cx * NG + i) * 2];
    s2 += ws[((long)cx * NG + i) * 2 + 1];
  }
  out[i * 2] = s1;
  out[i * 2 + 1] = s2;
}

__global__ void gn_bwd_finalize_dwdb_kernel(const float* __restrict__ slab,
                                            float* __restrict__ dw,
                                            float* __restrict__ db, int C,
                                            long rows) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= C) return;
  float a = 0.f, b = 0.f;
  for (long r = 0; r < rows; ++r) {
    a += slab[r * 2 * C + c];
    b += slab[r * 2 * C + C + c];
  }
  dw[c] = a;
  db[c] = b;
}

// --------------------------------------------------------------- apply fwd
template <typename T, typename WT, bool SILU, int V>
__global__ void gn_nhwc_apply_kernel(const T* __restrict__ x, const WT* __restrict__ w,
                                     const WT* __restrict__ b,
                                     const float* __restrict__ mean,
                                     const float* __restrict__ rstd,
                                     T* __restrict__ y, long total, int R, int C, int G) {
  const int Cg = C / G;
  const long nvec = total / V;
  const long RC = (long)R * C;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long e = i * V;
    long n = e / RC;
    int c = (int)(e % C);                  // C % V == 0: V consecutive c
    float xv[V], wv[V], bv[V], o[V];
    loadv<T, V>(x + e, xv);
    loadv<WT, V>(w + c, wv);
    loadv<WT, V>(b + c, bv);
#pragma unroll
    for (int k = 0; k < V; ++k) {
      int g = (c + k) / Cg;
      float m = mean[n * G + g];
      float rs = rstd[n * G + g];
      float z = (xv[k] - m) * rs * wv[k] + bv[k];
      o[k] = SILU ? silu(z) : z;
    }
    storev<T, V>(y + e, o);
  }
}

// --------------------------------------------------------------- stats bwd
// per-channel dw/db; per-group S1 = sum(w*dz), S2 = sum(w*dz*yhat)
template <typename T, typename WT, bool SILU, int V>
__global__ void gn_nhwc_bwd_stats_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                         const WT* __restrict__ w, const WT* __restrict__ b_,
                                         const float* __restrict__ mean,
                                         const float* __restrict__ rstd,
                                         float* __restrict__ ws,
                                         float* __restrict__ dwdb_slab,
                                         int N, int R, int C, int G, int rows_per_blk) {
  // smem [4*C]: per-channel S1/S2 (group sums) + per-channel dw/db
  // partials; everything leaves the block as PLAIN per-chunk slab
  // stores (no global atomics at all — deterministic, no zero-init;
  // the per-channel-atomic version measured 3.5-4.6x slower at V=8
  // from contention; finalize kernels reduce the slabs).
  // ws: [chunks, N, G, 2]; dwdb_slab: [chunks*N, 2C].
  extern __shared__ float smem[];
  float* sa = smem;                        // S1 per channel (w*dz)
  float* sb = smem + C;                    // S2 per channel (w*dz*yhat)
  float* sdw = smem + 2 * C;
  float* sdb = smem + 3 * C;
  const int n = blockIdx.y;
  const int r0 = blockIdx.x * rows_per_blk;
  const int r1 = min(R, r0 + rows_per_blk);
  const long base = (long)n * R * C;
  const int Cg = C / G;

  // (same wide/narrow split as the forward stats kernel)
  const bool wide = C >= (int)blockDim.x * V;
  int tpr = wide ? (int)blockDim.x : C / V;
  int rpar = wide ? 1 : (int)blockDim.x / tpr;
  int rgrp = wide ? 0 : (int)threadIdx.x / tpr;
  for (int c = threadIdx.x; c < 4 * C; c += blockDim.x) smem[c] = 0.f;
  __syncthreads();
  for (int c0 = (wide ? (int)threadIdx.x * V : ((int)threadIdx.x % tpr) * V);
       c0 < C; c0 += (wide ? (int)blockDim.x * V : C + 1)) {
    if (rgrp >= rpar) break;
    float m[V], rs[V], wc[V], bc[V];
    float a[V], bb[V], dwc[V], dbc[V];
#pragma unroll
    for (int k = 0; k < V; ++k) {
      const int g = (c0 + k) / Cg;
      m[k] = mean[(long)n * G + g];
      rs[k] = rstd[(long)n * G + g];
      wc[k] = to_f32<WT>(w[c0 + k]);
      bc[k] = to_f32<WT>(b_[c0 + k]);
      a[k] = bb[k] = dwc[k] = dbc[k] = 0.f;
    }
    for (int r = r0 + rgrp; r < r1; r += rpar) {
      long idx = base + (long)r * C + c0;
      float xv[V], gv[V];
      loadv<T, V>(x + idx, xv);
      loadv<T, V>(dy + idx, gv);
#pragma unroll
      for (int k = 0; k < V; ++k) {
        float yh = (xv[k] - m[k]) * rs[k];
        float dz = gv[k];
        if (SILU) dz *= dsilu(yh * wc[k] + bc[k]);
        float gx = dz * wc[k];
        a[k] += gx;
        bb[k] += gx * yh;
        dwc[k] += dz * yh;
        dbc[k] += dz;
      }
    }
#pragma unroll
    for (int k = 0; k < V; ++k) {
      if (wide) {
        sa[c0 + k] = a[k];
        sb[c0 + k] = bb[k];
        sdw[c0 + k] = dwc[k];
        sdb[c0 + k] = dbc[k];
      } else {
        atomicAdd(&sa[c0 + k], a[k]);
        atomicAdd(&sb[c0 + k], bb[k]);
        atomicAdd(&sdw[c0 + k], dwc[k]);
        atomicAdd(&sdb[c0 + k], dbc[k]);
      }
    }
  }
  __syncthreads();
  float* drow = dwdb_slab + ((long)blockIdx.x * N + n) * 2 * C;
  for (int c = threadIdx.x; c < C; c += blockDim.x) {
    drow[c] = sdw[c];
    drow[C + c] = sdb[c];
  }
  float* slot = ws + (((long)blockIdx.x * N + n) * G) * 2;
  for (int g = threadIdx.x; g < G; g += blockDim.x) {
    float a = 0.f, bb = 0.f;
    for (int c = g * Cg; c < (g + 1) * Cg; ++c) { a += sa[c]; bb += sb[c]; }
    slot[(long)g * 2] = a;
    slot[(long)g * 2 + 1] = bb;
  }
}

// --------------------------------------------------------------- apply bwd
template <typename T, typename WT, bool SILU, int V>
__global__ void gn_nhwc_bwd_apply_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                         const WT* __restrict__ w, const WT* __restrict__ b_,
                                         const float* __restrict__ mean,
                                         const float* __restrict__ rstd,
                                         const float* __restrict__ ws,
                                         T* __restrict__ dx, long total, int R, int C,
                                         int G, float inv_L) {
  const int Cg = C / G;
  const long nvec = total / V;
  const long RC = (long)R * C;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long e = i * V;
    long n = e / RC;
    int c = (int)(e % C);
    float xv[V], gv[V], wv[V], bv[V], o[V];
    loadv<T, V>(x + e, xv);
    loadv<T, V>(dy + e, gv);
    loadv<WT, V>(w + c, wv);
    loadv<WT, V>(b_ + c, bv);
#pragma unroll
    for (int k = 0; k < V; ++k) {
      int g = (c + k) / Cg;
      float m = mean[n * G + g];
      float rs = rstd[n * G + g];
      float m1 = ws[(n * G + g) * 2] * inv_L;
      float m2 = ws[(n * G + g) * 2 + 1] * inv_L;
      float yh = (xv[k] - m) * rs;
      float dz = gv[k];
      if (SILU) dz *= dsilu(yh * wv[k] + bv[k]);
      float gx = dz * wv[k];
      o[k] = rs * (gx - m1 - yh * m2);
    }
    storev<T, V>(dx + e, o);
  }
}

}  // namespace dcr_nhwc

// ==========================================================================
// Host launchers
// ==========================================================================
#include "dcr_launchers.h"

namespace dcr {

int gn_nhwc_chunks(int N, int R);  // forward decl for the binding

static inline int nhwc_row_chunks(int N, int R) {
  // target >= 512 workgroups to fill 256 CUs / 8 XCDs
  int chunks = (512 + N - 1) / N;
  if (chunks > R) chunks = R;
  if (chunks < 1) chunks = 1;
  return chunks;
}

template <typename T, typename WT, int V>
static void gn_nhwc_fwd_v(const void* x, const void* w, const void* b, void* y,
                          float* ws, float* mean, float* rstd, int N, int R,
                          int C, int G, float eps, bool silu, hipStream_t s) {
  int chunks = nhwc_row_chunks(N, R);
  int rows_per_blk = (R + chunks - 1) / chunks;
  dim3 grid(chunks, N), block(256);
  size_t lds = 2 * (size_t)C * sizeof(float);
  hipLaunchKernelGGL((dcr_nhwc::gn_nhwc_stats_kernel<T, V>), grid, block, lds, s,
                     (const T*)x, ws, N, R, C, G, rows_per_blk);
  long NG = (long)N * G;
  float inv_L = 1.f / ((float)R * (C / G));
  hipLaunchKernelGGL(dcr_nhwc::gn_finalize_kernel, dim3((NG + 255) / 256),
                     dim3(256), 0, s, ws, mean, rstd, NG, chunks, inv_L, eps);
  long total = (long)N * R * C;
  dim3 agrid((int)min((total / V + 255) / 256, (long)8192)), ablock(256);
  if (silu)
    hipLaunchKernelGGL((dcr_nhwc::gn_nhwc_apply_kernel<T, WT, true, V>), agrid, ablock, 0, s,
                       (const T*)x, (const WT*)w, (const WT*)b, mean, rstd, (T*)y, total, R, C, G);
  else
    hipLaunchKernelGGL((dcr_nhwc::gn_nhwc_apply_kernel<T, WT, false, V>), agrid, ablock, 0, s,
                       (const T*)x, (const WT*)w, (const WT*)b, mean, rstd, (T*)y, total, R, C, G);
}

template <typename T, typename WT>
static void gn_nhwc_fwd_t(const void* x, const void* w, const void* b, void* y,
                          float* ws, float* mean, float* rstd, int N, int R,
                          int C, int G, float eps, bool silu, hipStream_t s) {
  // 16 B/lane when the channel count admits it (f32 is already 16 B at V=4)
  if (C % 8 == 0 && sizeof(T) == 2)
    gn_nhwc_fwd_v<T, WT, 8>(x, w, b, y, ws, mean, rstd, N, R, C, G, eps, silu, s);
  else
    gn_nhwc_fwd_v<T, WT, 4>(x, w, b, y, ws, mean, rstd, N, R, C, G, eps, silu, s);
}

void gn_nhwc_fwd_launch(DType dt, const void* x, const void* w, const void* b,
                        bool w_f32, void* y, float* ws, float* mean, float* rstd,
                        int N, int R, int C, int G, float eps, bool silu,
                        hipStream_t s) {
  switch (dt) {
    case DT_F32: gn_nhwc_fwd_t<float, float>(x, w, b, y, ws, mean, rstd, N, R, C, G, eps, silu, s); break;
    case DT_F16:
      if (w_f32) gn_nhwc_fwd_t<__half, float>(x, w, b, y, ws, mean, rstd, N, R, C, G, eps, silu, s);
      else gn_nhwc_fwd_t<__half, __half>(x, w, b, y, ws, mean, rstd, N, R, C, G, eps, silu, s);
      break;
    case DT_BF16:
      if (w_f32) gn_nhwc_fwd_t<__hip_bfloat16, float>(x, w, b, y, ws, mean, rstd, N, R, C, G, eps, silu, s);
      else gn_nhwc_fwd_t<__hip_bfloat16, __hip_bfloat16>(x, w, b, y, ws, mean, rstd, N, R, C, G, eps, silu, s);
      break;
  }
}

template <typename T, typename WT, int V>
static void gn_nhwc_bwd_v(const void* dy, const void* x, const void* w,
                          const void* b, const float* mean, const float* rstd,
                          float* ws, void* dx, float* dw, float* db, int N,
                          int R, int C, int G, bool silu, hipStream_t s) {
  int chunks = nhwc_row_chunks(N, R);
  int rows_per_blk = (R + chunks - 1) / chunks;
  dim3 grid(chunks, N), block(256);
  size_t lds = 4 * (size_t)C * sizeof(float);
  // slab layout: ws_slab [chunks, N, G, 2] | dwdb_slab [chunks*N, 2C] |
  // finalized groups [N*G, 2] (the pointers are carved by the binding)
  long NG = (long)N * G;
  float* ws_slab = ws;
  float* dwdb_slab = ws + (long)chunks * NG * 2;
  float* ws_fin = dwdb_slab + (long)chunks * N * 2 * C;
  if (silu)
    hipLaunchKernelGGL((dcr_nhwc::gn_nhwc_bwd_stats_kernel<T, WT, true, V>), grid, block, lds, s,
                       (const T*)dy, (const T*)x, (const WT*)w, (const WT*)b, mean, rstd,
                       ws_slab, dwdb_slab, N, R, C, G, rows_per_blk);
  else
    hipLaunchKernelGGL((dcr_nhwc::gn_nhwc_bwd_stats_kernel<T, WT, false, V>), grid, block, lds, s,
                       (const T*)dy, (const T*)x, (const WT*)w, (const WT*)b, mean, rstd,
                       ws_slab, dwdb_slab, N, R, C, G, rows_per_blk);
  hipLaunchKernelGGL(dcr_nhwc::gn_bwd_finalize_groups_kernel,
                     dim3((NG + 255) / 256), dim3(256), 0, s, ws_slab, ws_fin,
                     NG, chunks);
  hipLaunchKernelGGL(dcr_nhwc::gn_bwd_finalize_dwdb_kernel,
                     dim3((C + 255) / 256), dim3(256), 0, s, dwdb_slab, dw, db,
                     C, (long)chunks * N);
  long total = (long)N * R * C;
  float inv_L = 1.f / ((float)R * (C / G));
  dim3 agrid((int)min((total / V + 255) / 256, (long)8192)), ablock(256);
  if (silu)
    hipLaunchKernelGGL((dcr_nhwc::gn_nhwc_bwd_apply_kernel<T, WT, true, V>), agrid, ablock, 0, s,
                       (const T*)dy, (const T*)x, (const WT*)w, (const WT*)b, mean, rstd, ws_fin, (T*)dx,
                       total, R, C, G, inv_L);
  else
    hipLaunchKernelGGL((dcr_nhwc::gn_nhwc_bwd_apply_kernel<T, WT, false, V>), agrid, ablock, 0, s,
                       (const T*)dy, (const T*)x, (const WT*)w, (const WT*)b, mean, rstd, ws_fin, (T*)dx,
                       total, R, C, G, inv_L);
}

template <typename T, typename WT>
static void gn_nhwc_bwd_t(const void* dy, const void* x, const void* w,
                          const void* b, const float* mean, const float* rstd,
                          float* ws, void* dx, float* dw, float* db, int N,
                          int R, int C, int G, bool silu, hipStream_t s) {
  if (C % 8 == 0 && sizeof(T) == 2)
    gn_nhwc_bwd_v<T, WT, 8>(dy, x, w, b, mean, rstd, ws, dx, dw, db, N, R, C, G, silu, s);
  else
    gn_nhwc_bwd_v<T, WT, 4>(dy, x, w, b, mean, rstd, ws, dx, dw, db, N, R, C, G, silu, s);
}

void gn_nhwc_bwd_launch(DType dt, const void* dy, const void* x, const void* w,
                        const void* b, bool w_f32, const float* mean,
                        const float* rstd, float* ws, void* dx, float* dw,
                        float* db, int N, int R, int C, int G, bool silu,
                        hipStream_t s) {
  switch (dt) {
    case DT_F32: gn_nhwc_bwd_t<float, float>(dy, x, w, b, mean, rstd, ws, dx, dw, db, N, R, C, G, silu, s); break;
    case DT_F16:
      if (w_f32) gn_nhwc_bwd_t<__half, float>(dy, x, w, b, mean, rstd, ws, dx, dw, db, N, R, C, G, silu, s);
      else gn_nhwc_bwd_t<__half, __half>(dy, x, w, b, mean, rstd, ws, dx, dw, db, N, R, C, G, silu, s);
      break;
    case DT_BF16:
      if (w_f32) gn_nhwc_bwd_t<__hip_bfloat16, float>(dy, x, w, b, mean, rstd, ws, dx, dw, db, N, R, C, G, silu, s);
      else gn_nhwc_bwd_t<__hip_bfloat16, __hip_bfloat16>(dy, x, w, b, mean, rstd, ws, dx, dw, db, N, R, C, G, silu, s);
      break;
  }
}

int gn_nhwc_chunks(int N, int R) { return nhwc_row_chunks(N, R); }

}  // namespace dcr
