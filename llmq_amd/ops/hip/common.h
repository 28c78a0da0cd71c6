// Common helpers for the llmq-amd CDNA4 (gfx950) kernels.
// Wave width is 64 on CDNA; never 32 (cdna_hip_programming.md §1).
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <hip/hip_fp8.h>

#include <type_traits>

#define WAVE 64
#define DEVINL __device__ __forceinline__

// ---- scalar conversions ---------------------------------------------------

template <typename T> DEVINL float to_f32(T x);
template <> DEVINL float to_f32<float>(float x) { return x; }
template <> DEVINL float to_f32<__hip_bfloat16>(__hip_bfloat16 x) {
  return __bfloat162float(x);
}
template <> DEVINL float to_f32<_Float16>(_Float16 x) { return (float)x; }
template <> DEVINL float to_f32<__hip_fp8_e4m3>(__hip_fp8_e4m3 x) {
  return static_cast<float>(x);
}

template <typename T> DEVINL T from_f32(float x);
template <> DEVINL float from_f32<float>(float x) { return x; }
template <> DEVINL __hip_bfloat16 from_f32<__hip_bfloat16>(float x) {
  return __float2bfloat16(x);
}
template <> DEVINL _Float16 from_f32<_Float16>(float x) { return (_Float16)x; }
template <> DEVINL __hip_fp8_e4m3 from_f32<__hip_fp8_e4m3>(float x) {
  // native v_cvt_pk_fp8_f32 (gfx950): one VOP instead of the library's
  // per-element software sequence (RNE, saturating — matches OCP e4m3fn)
  const int packed = __builtin_amdgcn_cvt_pk_fp8_f32(x, x, 0, false);
  __hip_fp8_e4m3 out;
  out.__x = (unsigned char)(packed & 0xff);
  return out;
}

// ---- vectorized 16-byte access (8 bf16 / 8 fp16 / 4 f32) -------------------
// G13: hipcc does not auto-vectorize bf16 loads; reinterpret as int4.

struct alignas(16) Bytes16 { int4 raw; };

template <typename T> struct Vec8 {
  // 8 elements of T when T is 2 bytes; 4 elements when 4 bytes.
  static constexpr int kElems = (sizeof(T) == 2) ? 8 : 4;
  T data[kElems];
};

template <typename T>
DEVINL Vec8<T> load16(const T* p) {
  Vec8<T> v;
  *reinterpret_cast<int4*>(v.data) = *reinterpret_cast<const int4*>(p);
  return v;
}

template <typename T>
DEVINL void store16(T* p, const Vec8<T>& v) {
  *reinterpret_cast<int4*>(p) = *reinterpret_cast<const int4*>(v.data);
}

template <typename T> struct Vec4 {
  // 8-byte vector: 4 elements of a 2-byte T, 2 of a 4-byte T.
  static constexpr int kElems = (sizeof(T) == 2) ? 4 : 2;
  T data[kElems];
};

template <typename T>
DEVINL Vec4<T> load8(const T* p) {
  Vec4<T> v;
  *reinterpret_cast<int2*>(v.data) = *reinterpret_cast<const int2*>(p);
  return v;
}

// ---- wave / group reductions ----------------------------------------------

DEVINL float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}

DEVINL float wave_reduce_max(float x) {
#pragma unroll
  for (int off = WAVE / 2; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_xor(x, off, WAVE));
  return x;
}

// Reduce across a 2^k-lane group (lanes with identical lane/width id get the sum).
template <int WIDTH>
DEVINL float group_reduce_sum(float x) {
#pragma unroll
  for (int off = WIDTH / 2; off > 0; off >>= 1) x += __shfl_xor(x, off, WAVE);
  return x;
}

// Block-level reduce via LDS: every thread contributes; thread 0's value is
// the result; all threads see it after the broadcast. `scratch` needs
// blockDim.x/WAVE floats.
DEVINL float block_reduce_sum(float x, float* scratch) {
  const int lane = threadIdx.x & (WAVE - 1);
  const int wid = threadIdx.x / WAVE;
  x = wave_reduce_sum(x);
  if (lane == 0) scratch[wid] = x;
  __syncthreads();
  const int nw = blockDim.x / WAVE;
  float r = (threadIdx.x < nw) ? scratch[threadIdx.x] : 0.0f;
  if (wid == 0) {
    r = wave_reduce_sum(r);
    if (lane == 0) scratch[0] = r;
  }
  __syncthreads();
  r = scratch[0];
  __syncthreads();
  return r;
}

DEVINL int ceil_div(int a, int b) { return (a + b - 1) / b; }
