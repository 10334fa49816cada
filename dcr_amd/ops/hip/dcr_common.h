// Common device helpers for dcr_amd CDNA4 (gfx950) kernels.
// Wave64 reductions, vectorized type-converting loads/stores, activation math.
// Written for MI355X: fp32 accumulation everywhere, 16B/lane loads where the
// layout permits (HBM3E-bound ops), LDS block reductions.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

#define DCR_WAVE 64

namespace dcr {

// ---------------------------------------------------------------- conversions
template <typename T> __device__ __forceinline__ float to_f32(T v);
template <> __device__ __forceinline__ float to_f32<float>(float v) { return v; }
template <> __device__ __forceinline__ float to_f32<__hip_bfloat16>(__hip_bfloat16 v) {
  return __bfloat162float(v);
}
template <> __device__ __forceinline__ float to_f32<__half>(__half v) {
  return __half2float(v);
}

template <typename T> __device__ __forceinline__ T from_f32(float v);
template <> __device__ __forceinline__ float from_f32<float>(float v) { return v; }
template <> __device__ __forceinline__ __hip_bfloat16 from_f32<__hip_bfloat16>(float v) {
  return __float2bfloat16(v);
}
template <> __device__ __forceinline__ __half from_f32<__half>(float v) {
  return __float2half(v);
}

// ------------------------------------------------------------- vec4 load/store
// One instruction per 4 elements: dwordx4 for f32, dwordx2 for bf16/f16.
struct f32x4 { float x, y, z, w; };

template <typename T>
__device__ __forceinline__ f32x4 load4(const T* __restrict__ p);

template <>
__device__ __forceinline__ f32x4 load4<float>(const float* __restrict__ p) {
  float4 v = *reinterpret_cast<const float4*>(p);
  return {v.x, v.y, v.z, v.w};
}

template <>
__device__ __forceinline__ f32x4 load4<__hip_bfloat16>(const __hip_bfloat16* __restrict__ p) {
  // 4 x bf16 = 8 bytes = one dwordx2
  ushort4 raw;
  *reinterpret_cast<uint2*>(&raw) = *reinterpret_cast<const uint2*>(p);
  f32x4 o;
  o.x = __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&raw.x));
  o.y = __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&raw.y));
  o.z = __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&raw.z));
  o.w = __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&raw.w));
  return o;
}

template <>
__device__ __forceinline__ f32x4 load4<__half>(const __half* __restrict__ p) {
  ushort4 raw;
  *reinterpret_cast<uint2*>(&raw) = *reinterpret_cast<const uint2*>(p);
  f32x4 o;
  o.x = __half2float(*reinterpret_cast<const __half*>(&raw.x));
  o.y = __half2float(*reinterpret_cast<const __half*>(&raw.y));
  o.z = __half2float(*reinterpret_cast<const __half*>(&raw.z));
  o.w = __half2float(*reinterpret_cast<const __half*>(&raw.w));
  return o;
}

template <typename T>
__device__ __forceinline__ void store4(T* __restrict__ p, f32x4 v);

template <>
__device__ __forceinline__ void store4<float>(float* __restrict__ p, f32x4 v) {
  *reinterpret_cast<float4*>(p) = make_float4(v.x, v.y, v.z, v.w);
}

template <>
__device__ __forceinline__ void store4<__hip_bfloat16>(__hip_bfloat16* __restrict__ p, f32x4 v) {
  ushort4 raw;
  __hip_bfloat16 a = __float2bfloat16(v.x), b = __float2bfloat16(v.y),
                 c = __float2bfloat16(v.z), d = __float2bfloat16(v.w);
  raw.x = *reinterpret_cast<unsigned short*>(&a);
  raw.y = *reinterpret_cast<unsigned short*>(&b);
  raw.z = *reinterpret_cast<unsigned short*>(&c);
  raw.w = *reinterpret_cast<unsigned short*>(&d);
  *reinterpret_cast<uint2*>(p) = *reinterpret_cast<uint2*>(&raw);
}

template <>
__device__ __forceinline__ void store4<__half>(__half* __restrict__ p, f32x4 v) {
  ushort4 raw;
  __half a = __float2half(v.x), b = __float2half(v.y),
         c = __float2half(v.z), d = __float2half(v.w);
  raw.x = *reinterpret_cast<unsigned short*>(&a);
  raw.y = *reinterpret_cast<unsigned short*>(&b);
  raw.z = *reinterpret_cast<unsigned short*>(&c);
  raw.w = *reinterpret_cast<unsigned short*>(&d);
  *reinterpret_cast<uint2*>(p) = *reinterpret_cast<uint2*>(&raw);
}

// ------------------------------------------------------------- vec8 load/store
// 8 elements per instruction: 16 B/lane for bf16/f16 — the HBM3E
// coalescing sweet spot (guide G13); f32 uses two dwordx4.
struct f32x8 { f32x4 lo, hi; };

template <typename T>
__device__ __forceinline__ f32x8 load8(const T* __restrict__ p);

template <>
__device__ __forceinline__ f32x8 load8<float>(const float* __restrict__ p) {
  return {load4<float>(p), load4<float>(p + 4)};
}

template <>
__device__ __forceinline__ f32x8 load8<__hip_bfloat16>(const __hip_bfloat16* __restrict__ p) {
  ushort raw[8];
  *reinterpret_cast<uint4*>(raw) = *reinterpret_cast<const uint4*>(p);
  f32x8 o;
#pragma unroll
  for (int k = 0; k < 4; ++k) {
    (&o.lo.x)[k] = __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&raw[k]));
    (&o.hi.x)[k] = __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&raw[4 + k]));
  }
  return o;
}

template <>
__device__ __forceinline__ f32x8 load8<__half>(const __half* __restrict__ p) {
  ushort raw[8];
  *reinterpret_cast<uint4*>(raw) = *reinterpret_cast<const uint4*>(p);
  f32x8 o;
#pragma unroll
  for (int k = 0; k < 4; ++k) {
    (&o.lo.x)[k] = __half2float(*reinterpret_cast<const __half*>(&raw[k]));
    (&o.hi.x)[k] = __half2float(*reinterpret_cast<const __half*>(&raw[4 + k]));
  }
  return o;
}

template <typename T>
__device__ __forceinline__ void store8(T* __restrict__ p, f32x8 v);

template <>
__device__ __forceinline__ void store8<float>(float* __restrict__ p, f32x8 v) {
  store4<float>(p, v.lo);
  store4<float>(p + 4, v.hi);
}

template <>
__device__ __forceinline__ void store8<__hip_bfloat16>(__hip_bfloat16* __restrict__ p, f32x8 v) {
  ushort raw[8];
#pragma unroll
  for (int k = 0; k < 4; ++k) {
    __hip_bfloat16 a = __float2bfloat16((&v.lo.x)[k]);
    __hip_bfloat16 b = __float2bfloat16((&v.hi.x)[k]);
    raw[k] = *reinterpret_cast<unsigned short*>(&a);
    raw[4 + k] = *reinterpret_cast<unsigned short*>(&b);
  }
  *reinterpret_cast<uint4*>(p) = *reinterpret_cast<uint4*>(raw);
}

template <>
__device__ __forceinline__ void store8<__half>(__half* __restrict__ p, f32x8 v) {
  ushort raw[8];
#pragma unroll
  for (int k = 0; k < 4; ++k) {
    __half a = __float2half((&v.lo.x)[k]);
    __half b = __float2half((&v.hi.x)[k]);
    raw[k] = *reinterpret_cast<unsigned short*>(&a);
    raw[4 + k] = *reinterpret_cast<unsigned short*>(&b);
  }
  *reinterpret_cast<uint4*>(p) = *reinterpret_cast<uint4*>(raw);
}

// --------------------------------------------------------------- reductions
__device__ __forceinline__ float wave_reduce_sum(float v) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) v += __shfl_xor(v, off, DCR_WAVE);
  return v;
}

__device__ __forceinline__ float2 wave_reduce_sum2(float a, float b) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) {
    a += __shfl_xor(a, off, DCR_WAVE);
    b += __shfl_xor(b, off, DCR_WAVE);
  }
  return make_float2(a, b);
}

// Block reduction of two values; `lds` must hold 2*(blockDim.x/64) floats.
// Result valid in ALL threads.
__device__ __forceinline__ float2 block_reduce_sum2(float a, float b, float* lds) {
  const int wid = threadIdx.x / DCR_WAVE;
  const int lane = threadIdx.x % DCR_WAVE;
  const int nw = blockDim.x / DCR_WAVE;
  float2 w = wave_reduce_sum2(a, b);
  if (lane == 0) { lds[2 * wid] = w.x; lds[2 * wid + 1] = w.y; }
  __syncthreads();
  float ra = 0.f, rb = 0.f;
#pragma unroll 1
  for (int i = 0; i < nw; ++i) { ra += lds[2 * i]; rb += lds[2 * i + 1]; }
  __syncthreads();
  return make_float2(ra, rb);
}

// --------------------------------------------------------------- activations
__device__ __forceinline__ float silu(float z) { return z / (1.f + __expf(-z)); }

__device__ __forceinline__ float dsilu(float z) {
  float s = 1.f / (1.f + __expf(-z));
  return s * (1.f + z * (1.f - s));
}

__device__ __forceinline__ float gelu_erf(float g) {
  return 0.5f * g * (1.f + erff(g * 0.70710678118654752f));
}

__device__ __forceinline__ float dgelu_erf(float g) {
  // d/dg [g * Phi(g)] = Phi(g) + g * phi(g)
  float Phi = 0.5f * (1.f + erff(g * 0.70710678118654752f));
  float phi = 0.3989422804014327f * __expf(-0.5f * g * g);
  return Phi + g * phi;
}

}  // namespace dcr
