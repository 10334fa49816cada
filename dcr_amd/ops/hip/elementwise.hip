// Elementwise fused kernels for MI355X: GEGLU fwd/bwd, fused flat AdamW,
// DDPM scheduler math (add_noise / velocity), CFG combine.
// All grid-stride, vec4 (dwordx4 f32 / dwordx2 bf16), fp32 math.
// Reference ops these replace: SURVEY.md §2.4.D/E.

#include "dcr_common.h"

using namespace dcr;

// ---------------------------------------------------------------- GEGLU
// x: [M, 2N] -> y: [M, N];  y = a * gelu(g), a = x[:, :N], g = x[:, N:]
template <typename T>
__global__ void geglu_fwd_kernel(const T* __restrict__ x, T* __restrict__ y,
                                 long M, long N) {
  const long nvec = M * N / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long e = i * 4;
    long row = e / N, col = e % N;
    const T* base = x + row * 2 * N;
    f32x4 a = load4<T>(base + col);
    f32x4 g = load4<T>(base + N + col);
    f32x4 o;
    o.x = a.x * gelu_erf(g.x);
    o.y = a.y * gelu_erf(g.y);
    o.z = a.z * gelu_erf(g.z);
    o.w = a.w * gelu_erf(g.w);
    store4<T>(y + row * N + col, o);
  }
}

template <typename T>
__global__ void geglu_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                 T* __restrict__ dx, long M, long N) {
  const long nvec = M * N / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long e = i * 4;
    long row = e / N, col = e % N;
    const T* base = x + row * 2 * N;
    T* dbase = dx + row * 2 * N;
    f32x4 a = load4<T>(base + col);
    f32x4 g = load4<T>(base + N + col);
    f32x4 d = load4<T>(dy + row * N + col);
    f32x4 da, dg;
    da.x = d.x * gelu_erf(g.x); dg.x = d.x * a.x * dgelu_erf(g.x);
    da.y = d.y * gelu_erf(g.y); dg.y = d.y * a.y * dgelu_erf(g.y);
    da.z = d.z * gelu_erf(g.z); dg.z = d.z * a.z * dgelu_erf(g.z);
    da.w = d.w * gelu_erf(g.w); dg.w = d.w * a.w * dgelu_erf(g.w);
    store4<T>(dbase + col, da);
    store4<T>(dbase + N + col, dg);
  }
}

// ---------------------------------------------------------------- AdamW
// One kernel over the flat fp32 param/grad/m/v arenas (FusedAdamW).
__global__ void adamw_kernel(float* __restrict__ p, const float* __restrict__ g,
                             float* __restrict__ m, float* __restrict__ v,
                             long n, float lr, float beta1, float beta2,
                             float eps, float wd, float inv_bc1, float inv_bc2) {
  const long nvec = n / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long e = i * 4;
    f32x4 pv = load4<float>(p + e);
    f32x4 gv = load4<float>(g + e);
    f32x4 mv = load4<float>(m + e);
    f32x4 vv = load4<float>(v + e);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float gg = (&gv.x)[k];
      float mm = beta1 * (&mv.x)[k] + (1.f - beta1) * gg;
      float vvk = beta2 * (&vv.x)[k] + (1.f - beta2) * gg * gg;
      (&mv.x)[k] = mm;
      (&vv.x)[k] = vvk;
      float denom = sqrtf(vvk * inv_bc2) + eps;
      float upd = (mm * inv_bc1) / denom + wd * (&pv.x)[k];
      (&pv.x)[k] -= lr * upd;
    }
    store4<float>(p + e, pv);
    store4<float>(m + e, mv);
    store4<float>(v + e, vv);
  }
  // scalar tail
  long tail = nvec * 4 + (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (tail < n) {
    float gg = g[tail];
    float mm = beta1 * m[tail] + (1.f - beta1) * gg;
    float vvk = beta2 * v[tail] + (1.f - beta2) * gg * gg;
    m[tail] = mm; v[tail] = vvk;
    float denom = sqrtf(vvk * inv_bc2) + eps;
    p[tail] -= lr * ((mm * inv_bc1) / denom + wd * p[tail]);
  }
}

// AdamW with bf16 model params + fp32 master (pure-bf16 training mode):
// reads bf16 grads, updates fp32 master + Adam state, writes bf16 params.
// Removes the autocast weight-cast traffic (profiles/r01: ~10 ms/step of
// bf16<->f32 copies) and halves DDP gradient bytes over xGMI.
__global__ void adamw_bf16_kernel(__hip_bfloat16* __restrict__ p,
                                  const __hip_bfloat16* __restrict__ g,
                                  float* __restrict__ master,
                                  float* __restrict__ m, float* __restrict__ v,
                                  long n, float lr, float beta1, float beta2,
                                  float eps, float wd, float inv_bc1,
                                  float inv_bc2) {
  const long nvec = n / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long e = i * 4;
    f32x4 gv = load4<__hip_bfloat16>(g + e);
    f32x4 pv = load4<float>(master + e);
    f32x4 mv = load4<float>(m + e);
    f32x4 vv = load4<float>(v + e);
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      float gg = (&gv.x)[kk];
      float mm = beta1 * (&mv.x)[kk] + (1.f - beta1) * gg;
      float vvk = beta2 * (&vv.x)[kk] + (1.f - beta2) * gg * gg;
      (&mv.x)[kk] = mm;
      (&vv.x)[kk] = vvk;
      float denom = sqrtf(vvk * inv_bc2) + eps;
      (&pv.x)[kk] -= lr * ((mm * inv_bc1) / denom + wd * (&pv.x)[kk]);
    }
    store4<float>(master + e, pv);
    store4<float>(m + e, mv);
    store4<float>(v + e, vv);
    store4<__hip_bfloat16>(p + e, pv);
  }
  long tail = nvec * 4 + (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (tail < n) {
    float gg = to_f32<__hip_bfloat16>(g[tail]);
    float mm = beta1 * m[tail] + (1.f - beta1) * gg;
    float vvk = beta2 * v[tail] + (1.f - beta2) * gg * gg;
    m[tail] = mm; v[tail] = vvk;
    float denom = sqrtf(vvk * inv_bc2) + eps;
    float pv = master[tail] - lr * ((mm * inv_bc1) / denom + wd * master[tail]);
    master[tail] = pv;
    p[tail] = from_f32<__hip_bfloat16>(pv);
  }
}

// ------------------------------------------------- DDPM add_noise / velocity
// out = sa * A + sb * B, with (sa, sb) = f(alphas_cumprod[t[batch]]).
// MODE 0: add_noise  (sa=sqrt(ac), sb=sqrt(1-ac), A=x0, B=noise)
// MODE 1: velocity   (sa=sqrt(ac), sb=-sqrt(1-ac), A=noise, B=x0)
template <typename T, int MODE>
__global__ void sched_kernel(const T* __restrict__ x0, const T* __restrict__ noise,
                             const float* __restrict__ ac, const long* __restrict__ t,
                             T* __restrict__ out, long per_sample, long total) {
  const long nvec = total / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long e = i * 4;
    long bidx = e / per_sample;  // per_sample % 4 == 0 guaranteed by binding
    float a = ac[t[bidx]];
    float sa = sqrtf(a), sb = sqrtf(1.f - a);
    f32x4 xa = load4<T>(x0 + e);
    f32x4 xb = load4<T>(noise + e);
    f32x4 o;
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      if (MODE == 0)
        (&o.x)[k] = sa * (&xa.x)[k] + sb * (&xb.x)[k];
      else
        (&o.x)[k] = sa * (&xb.x)[k] - sb * (&xa.x)[k];
    }
    store4<T>(out + e, o);
  }
}

// ------------------------------------------------------- sampler lincomb
// out = a*X + b*Y (+ c*Z). Every DDIM / DPM-Solver++ update is one such
// fused elementwise op with host-side scalar coefficients — the whole
// sampling step is ONE kernel (north star: scheduler math as CDNA4 HIP).
template <typename T, bool HASZ>
__global__ void lincomb_kernel(const T* __restrict__ X, const T* __restrict__ Y,
                               const T* __restrict__ Z, T* __restrict__ out,
                               float a, float b, float c, long total) {
  const long nvec = total / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long e = i * 4;
    f32x4 xv = load4<T>(X + e);
    f32x4 yv = load4<T>(Y + e);
    f32x4 o;
#pragma unroll
    for (int k = 0; k < 4; ++k)
      (&o.x)[k] = a * (&xv.x)[k] + b * (&yv.x)[k];
    if (HASZ) {
      f32x4 zv = load4<T>(Z + e);
#pragma unroll
      for (int k = 0; k < 4; ++k)
        (&o.x)[k] += c * (&zv.x)[k];
    }
    store4<T>(out + e, o);
  }
}

// ---------------------------------------------------------------- CFG
template <typename T>
__global__ void cfg_kernel(const T* __restrict__ eu, const T* __restrict__ et,
                           T* __restrict__ out, float s, long total) {
  const long nvec = total / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long e = i * 4;
    f32x4 a = load4<T>(eu + e);
    f32x4 b = load4<T>(et + e);
    f32x4 o;
    o.x = a.x + s * (b.x - a.x);
    o.y = a.y + s * (b.y - a.y);
    o.z = a.z + s * (b.z - a.z);
    o.w = a.w + s * (b.w - a.w);
    store4<T>(out + e, o);
  }
}

#define DCR_INST_EW(T)                                                            \
  template __global__ void geglu_fwd_kernel<T>(const T*, T*, long, long);         \
  template __global__ void geglu_bwd_kernel<T>(const T*, const T*, T*, long, long); \
  template __global__ void sched_kernel<T, 0>(const T*, const T*, const float*,   \
      const long*, T*, long, long);                                               \
  template __global__ void sched_kernel<T, 1>(const T*, const T*, const float*,   \
      const long*, T*, long, long);                                               \
  template __global__ void cfg_kernel<T>(const T*, const T*, T*, float, long); \
  template __global__ void lincomb_kernel<T, true>(const T*, const T*, const T*, \
      T*, float, float, float, long);                                          \
  template __global__ void lincomb_kernel<T, false>(const T*, const T*, const T*, \
      T*, float, float, float, long);

DCR_INST_EW(float)
DCR_INST_EW(__hip_bfloat16)
DCR_INST_EW(__half)

// ===========================================================================
// Host launchers
// ===========================================================================
#include "dcr_launchers.h"

// ---------------------------------------------------------------------------
// Device-state AdamW (round-2 draft): the whole optimizer step — grad-norm,
// clip coefficient, bias-correction powers, update — reads its scalars from
// device memory so the 3-kernel sequence can be captured ONCE into a
// hipGraph and replayed with zero host work per step (the LR schedule
// writes hyper[0] before replay; everything else evolves on device).
// hyper layout (float[8]):
//   0 lr | 1 beta1^t | 2 beta2^t | 3 inv_bc1 | 4 inv_bc2
//   5 clip_coef | 6 grad_norm_sq accumulator | 7 step count
// init: {lr, 1, 1, 1, 1, 1, 0, 0}.
// NOT dispatched this round — FusedAdamW keeps the host-scalar path;
// hardware validation + trainer wiring is round 2 (DCR_DEV_ADAMW=1 tests).
// ---------------------------------------------------------------------------
template <typename T>
__global__ void gnormsq_kernel(const T* __restrict__ g, long n,
                               float* __restrict__ accum) {
  __shared__ float lds[2 * 256 / 64];
  float s = 0.f;
  const long nvec = n / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    f32x4 gv = load4<T>(g + i * 4);
    s += gv.x * gv.x + gv.y * gv.y + gv.z * gv.z + gv.w * gv.w;
  }
  long tail = nvec * 4 + (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (tail < n) {
    float x = to_f32<T>(g[tail]);
    s += x * x;
  }
  float2 r = block_reduce_sum2(s, 0.f, lds);
  if (threadIdx.x == 0) atomicAdd(accum, r.x);
}

__global__ void adamw_prologue_kernel(float* __restrict__ hyper, float beta1,
                                      float beta2, float max_norm) {
  // one thread: advance step state, turn the norm accumulator into the
  // clip coefficient, reset the accumulator for the next step
  float b1t = hyper[1] * beta1;
  float b2t = hyper[2] * beta2;
  hyper[1] = b1t;
  hyper[2] = b2t;
  hyper[3] = 1.f / (1.f - b1t);
  hyper[4] = 1.f / (1.f - b2t);
  float gn = sqrtf(hyper[6]);
  hyper[5] = (max_norm > 0.f && gn > max_norm) ? max_norm / (gn + 1e-6f) : 1.f;
  hyper[6] = 0.f;
  hyper[7] += 1.f;
}

__global__ void adamw_dev_kernel(float* __restrict__ p, const float* __restrict__ g,
                                 float* __restrict__ m, float* __restrict__ v,
                                 long n, float beta1, float beta2, float eps,
                                 float wd, const float* __restrict__ hyper) {
  const float lr = hyper[0];
  const float inv_bc1 = hyper[3];
  const float inv_bc2 = hyper[4];
  const float clip = hyper[5];
  const long nvec = n / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long e = i * 4;
    f32x4 pv = load4<float>(p + e);
    f32x4 gv = load4<float>(g + e);
    f32x4 mv = load4<float>(m + e);
    f32x4 vv = load4<float>(v + e);
#pragma unroll
    for (int k = 0; k < 4; ++k) {
      float gg = (&gv.x)[k] * clip;
      float mm = beta1 * (&mv.x)[k] + (1.f - beta1) * gg;
      float vvk = beta2 * (&vv.x)[k] + (1.f - beta2) * gg * gg;
      (&mv.x)[k] = mm;
      (&vv.x)[k] = vvk;
      float denom = sqrtf(vvk * inv_bc2) + eps;
      (&pv.x)[k] -= lr * ((mm * inv_bc1) / denom + wd * (&pv.x)[k]);
    }
    store4<float>(p + e, pv);
    store4<float>(m + e, mv);
    store4<float>(v + e, vv);
  }
  long tail = nvec * 4 + (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (tail < n) {
    float gg = g[tail] * clip;
    float mm = beta1 * m[tail] + (1.f - beta1) * gg;
    float vvk = beta2 * v[tail] + (1.f - beta2) * gg * gg;
    m[tail] = mm; v[tail] = vvk;
    float denom = sqrtf(vvk * inv_bc2) + eps;
    p[tail] -= lr * ((mm * inv_bc1) / denom + wd * p[tail]);
  }
}

__global__ void adamw_bf16_dev_kernel(__hip_bfloat16* __restrict__ p,
                                      const __hip_bfloat16* __restrict__ g,
                                      float* __restrict__ master,
                                      float* __restrict__ m, float* __restrict__ v,
                                      long n, float beta1, float beta2, float eps,
                                      float wd, const float* __restrict__ hyper) {
  const float lr = hyper[0];
  const float inv_bc1 = hyper[3];
  const float inv_bc2 = hyper[4];
  const float clip = hyper[5];
  const long nvec = n / 4;
  for (long i = (long)blockIdx.x * blockDim.x + threadIdx.x; i < nvec;
       i += (long)gridDim.x * blockDim.x) {
    long e = i * 4;
    f32x4 gv = load4<__hip_bfloat16>(g + e);
    f32x4 pv = load4<float>(master + e);
    f32x4 mv = load4<float>(m + e);
    f32x4 vv = load4<float>(v + e);
#pragma unroll
    for (int kk = 0; kk < 4; ++kk) {
      float gg = (&gv.x)[kk] * clip;
      float mm = beta1 * (&mv.x)[kk] + (1.f - beta1) * gg;
      float vvk = beta2 * (&vv.x)[kk] + (1.f - beta2) * gg * gg;
      (&mv.x)[kk] = mm;
      (&vv.x)[kk] = vvk;
      float denom = sqrtf(vvk * inv_bc2) + eps;
      (&pv.x)[kk] -= lr * ((mm * inv_bc1) / denom + wd * (&pv.x)[kk]);
    }
    store4<float>(master + e, pv);
    store4<float>(m + e, mv);
    store4<float>(v + e, vv);
    store4<__hip_bfloat16>(p + e, pv);
  }
  long tail = nvec * 4 + (long)blockIdx.x * blockDim.x + threadIdx.x;
  if (tail < n) {
    float gg = to_f32<__hip_bfloat16>(g[tail]) * clip;
    float mm = beta1 * m[tail] + (1.f - beta1) * gg;
    float vvk = beta2 * v[tail] + (1.f - beta2) * gg * gg;
    m[tail] = mm; v[tail] = vvk;
    float denom = sqrtf(vvk * inv_bc2) + eps;
    float pv = master[tail] - lr * ((mm * inv_bc1) / denom + wd * master[tail]);
    master[tail] = pv;
    p[tail] = from_f32<__hip_bfloat16>(pv);
  }
}

namespace dcr {

static inline int ew_blocks(long nvec, int block = 256) {
  long b = (nvec + block - 1) / block;
  if (b > 8192) b = 8192;   // grid-stride covers the rest; >> 256 WGs fills 8 XCDs
  if (b < 1) b = 1;
  return (int)b;
}

void geglu_fwd_launch(DType dt, const void* x, void* y, long M, long N,
                      hipStream_t s) {
  dim3 grid(ew_blocks(M * N / 4)), block(256);
  switch (dt) {
    case DT_F32: hipLaunchKernelGGL((geglu_fwd_kernel<float>), grid, block, 0, s, (const float*)x, (float*)y, M, N); break;
    case DT_F16: hipLaunchKernelGGL((geglu_fwd_kernel<__half>), grid, block, 0, s, (const __half*)x, (__half*)y, M, N); break;
    case DT_BF16: hipLaunchKernelGGL((geglu_fwd_kernel<__hip_bfloat16>), grid, block, 0, s, (const __hip_bfloat16*)x, (__hip_bfloat16*)y, M, N); break;
  }
}

void geglu_bwd_launch(DType dt, const void* dy, const void* x, void* dx,
                      long M, long N, hipStream_t s) {
  dim3 grid(ew_blocks(M * N / 4)), block(256);
  switch (dt) {
    case DT_F32: hipLaunchKernelGGL((geglu_bwd_kernel<float>), grid, block, 0, s, (const float*)dy, (const float*)x, (float*)dx, M, N); break;
    case DT_F16: hipLaunchKernelGGL((geglu_bwd_kernel<__half>), grid, block, 0, s, (const __half*)dy, (const __half*)x, (__half*)dx, M, N); break;
    case DT_BF16: hipLaunchKernelGGL((geglu_bwd_kernel<__hip_bfloat16>), grid, block, 0, s, (const __hip_bfloat16*)dy, (const __hip_bfloat16*)x, (__hip_bfloat16*)dx, M, N); break;
  }
}

void adamw_launch(float* p, const float* g, float* m, float* v, long n,
                  float lr, float b1, float b2, float eps, float wd, long step,
                  hipStream_t s) {
  float bc1 = 1.f - powf(b1, (float)step);
  float bc2 = 1.f - powf(b2, (float)step);
  dim3 grid(ew_blocks(n / 4)), block(256);
  hipLaunchKernelGGL(adamw_kernel, grid, block, 0, s, p, g, m, v, n, lr, b1, b2,
                     eps, wd, 1.f / bc1, 1.f / bc2);
}

void adamw_bf16_launch(void* p, const void* g, float* master, float* m,
                       float* v, long n, float lr, float b1, float b2,
                       float eps, float wd, long step, hipStream_t s) {
  float bc1 = 1.f - powf(b1, (float)step);
  float bc2 = 1.f - powf(b2, (float)step);
  dim3 grid(ew_blocks(n / 4)), block(256);
  hipLaunchKernelGGL(adamw_bf16_kernel, grid, block, 0, s,
                     (__hip_bfloat16*)p, (const __hip_bfloat16*)g, master, m, v,
                     n, lr, b1, b2, eps, wd, 1.f / bc1, 1.f / bc2);
}

// device-state AdamW (round-2 draft): norm -> prologue -> update on one
// stream; every scalar that changes per step lives in hyper[8] on device
void adamw_dev_launch(int bf16, void* p, const void* g, float* master,
                      float* m, float* v, long n, float b1, float b2,
                      float eps, float wd, float max_norm, float* hyper,
                      hipStream_t s) {
  dim3 grid(ew_blocks(n / 4)), block(256);
  if (bf16)
    hipLaunchKernelGGL((gnormsq_kernel<__hip_bfloat16>), grid, block, 0, s,
                       (const __hip_bfloat16*)g, n, hyper + 6);
  else
    hipLaunchKernelGGL((gnormsq_kernel<float>), grid, block, 0, s,
                       (const float*)g, n, hyper + 6);
  hipLaunchKernelGGL(adamw_prologue_kernel, dim3(1), dim3(1), 0, s, hyper,
                     b1, b2, max_norm);
  if (bf16)
    hipLaunchKernelGGL(adamw_bf16_dev_kernel, grid, block, 0, s,
                       (__hip_bfloat16*)p, (const __hip_bfloat16*)g, master,
                       m, v, n, b1, b2, eps, wd, hyper);
  else
    hipLaunchKernelGGL(adamw_dev_kernel, grid, block, 0, s, (float*)p,
                       (const float*)g, m, v, n, b1, b2, eps, wd, hyper);
}

void sched_launch(DType dt, int mode, const void* x0, const void* noise,
                  const float* ac, const long* t, void* out, long per_sample,
                  long total, hipStream_t s) {
  dim3 grid(ew_blocks(total / 4)), block(256);
#define SCHED_CASE(T)                                                              \
  if (mode == 0)                                                                   \
    hipLaunchKernelGGL((sched_kernel<T, 0>), grid, block, 0, s, (const T*)x0,      \
                       (const T*)noise, ac, t, (T*)out, per_sample, total);        \
  else                                                                             \
    hipLaunchKernelGGL((sched_kernel<T, 1>), grid, block, 0, s, (const T*)x0,      \
                       (const T*)noise, ac, t, (T*)out, per_sample, total);
  switch (dt) {
    case DT_F32: { SCHED_CASE(float) break; }
    case DT_F16: { SCHED_CASE(__half) break; }
    case DT_BF16: { SCHED_CASE(__hip_bfloat16) break; }
  }
#undef SCHED_CASE
}

void lincomb_launch(DType dt, const void* X, const void* Y, const void* Z,
                    void* out, float a, float b, float c, long total,
                    hipStream_t s) {
  dim3 grid(ew_blocks(total / 4)), block(256);
#define LC_CASE(T)                                                                \
  if (Z)                                                                          \
    hipLaunchKernelGGL((lincomb_kernel<T, true>), grid, block, 0, s,              \
                       (const T*)X, (const T*)Y, (const T*)Z, (T*)out, a, b, c,   \
                       total);                                                    \
  else                                                                            \
    hipLaunchKernelGGL((lincomb_kernel<T, false>), grid, block, 0, s,             \
                       (const T*)X, (const T*)Y, (const T*)Z, (T*)out, a, b, c,   \
                       total);
  switch (dt) {
    case DT_F32: { LC_CASE(float) break; }
    case DT_F16: { LC_CASE(__half) break; }
    case DT_BF16: { LC_CASE(__hip_bfloat16) break; }
  }
#undef LC_CASE
}

void cfg_launch(DType dt, const void* eu, const void* et, void* out, float s_,
                long total, hipStream_t s) {
  dim3 grid(ew_blocks(total / 4)), block(256);
  switch (dt) {
    case DT_F32: hipLaunchKernelGGL((cfg_kernel<float>), grid, block, 0, s, (const float*)eu, (const float*)et, (float*)out, s_, total); break;
    case DT_F16: hipLaunchKernelGGL((cfg_kernel<__half>), grid, block, 0, s, (const __half*)eu, (const __half*)et, (__half*)out, s_, total); break;
    case DT_BF16: hipLaunchKernelGGL((cfg_kernel<__hip_bfloat16>), grid, block, 0, s, (const __hip_bfloat16*)eu, (const __hip_bfloat16*)et, (__hip_bfloat16*)out, s_, total); break;
  }
}

}  // namespace dcr
