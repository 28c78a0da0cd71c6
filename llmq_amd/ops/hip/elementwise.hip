// Fused elementwise / normalisation kernels (gfx950).
//
// All memory-bound: the design rule is vectorized 16-byte accesses per lane
// (cdna_hip_programming.md G13: scalar bf16 loads are 2-2.5x slower) and a
// single pass per tensor (fused residual-add + RMSNorm writes both outputs
// from one read of each input). fp32 accumulation throughout.

#include "common.h"

// ---------------------------------------------------------------- RMSNorm --
// One workgroup per row. 256 threads, each handling 16B chunks grid-stride.
// `offset` = 1.0 for Gemma's (1 + w) convention.

template <typename T>
__global__ void rmsnorm_kernel(T* __restrict__ out, const T* __restrict__ in,
                               const T* __restrict__ weight, int hidden,
                               float eps, float offset) {
  constexpr int VE = Vec8<T>::kElems;
  __shared__ float red[4];
  const int row = blockIdx.x;
  const T* x = in + (long)row * hidden;
  T* o = out + (long)row * hidden;

  float ss = 0.f;
  for (int i = threadIdx.x * VE; i < hidden; i += blockDim.x * VE) {
    Vec8<T> v = load16(x + i);
#pragma unroll
    for (int j = 0; j < VE; ++j) {
      float f = to_f32(v.data[j]);
      ss += f * f;
    }
  }
  ss = block_reduce_sum(ss, red);
  const float inv = rsqrtf(ss / hidden + eps);

  for (int i = threadIdx.x * VE; i < hidden; i += blockDim.x * VE) {
    Vec8<T> v = load16(x + i);
    Vec8<T> w = load16(weight + i);
    Vec8<T> r;
#pragma unroll
    for (int j = 0; j < VE; ++j)
      r.data[j] = from_f32<T>(to_f32(v.data[j]) * inv * (to_f32(w.data[j]) + offset));
    store16(o + i, r);
  }
}

// x = rmsnorm(residual + x); residual = residual + x_in. In-place on both.
template <typename T>
__global__ void fused_add_rmsnorm_kernel(T* __restrict__ x, T* __restrict__ residual,
                                         const T* __restrict__ weight, int hidden,
                                         float eps, float offset) {
  constexpr int VE = Vec8<T>::kElems;
  __shared__ float red[4];
  const int row = blockIdx.x;
  T* xr = x + (long)row * hidden;
  T* rr = residual + (long)row * hidden;

  float ss = 0.f;
  for (int i = threadIdx.x * VE; i < hidden; i += blockDim.x * VE) {
    Vec8<T> vx = load16(xr + i);
    Vec8<T> vr = load16(rr + i);
    Vec8<T> sum;
#pragma unroll
    for (int j = 0; j < VE; ++j) {
      float f = to_f32(vx.data[j]) + to_f32(vr.data[j]);
      sum.data[j] = from_f32<T>(f);
      f = to_f32(sum.data[j]);  // accumulate on the ROUNDED value (matches ref)
      ss += f * f;
    }
    store16(rr + i, sum);  // new residual
  }
  ss = block_reduce_sum(ss, red);
  const float inv = rsqrtf(ss / hidden + eps);

  for (int i = threadIdx.x * VE; i < hidden; i += blockDim.x * VE) {
    Vec8<T> v = load16(rr + i);
    Vec8<T> w = load16(weight + i);
    Vec8<T> r;
#pragma unroll
    for (int j = 0; j < VE; ++j)
      r.data[j] = from_f32<T>(to_f32(v.data[j]) * inv * (to_f32(w.data[j]) + offset));
    store16(xr + i, r);
  }
}

// ------------------------------------------------------------ activations --
// in: [rows, 2*d] (gate || up) -> out: [rows, d]

template <typename T, bool GELU>
__global__ void act_and_mul_kernel(T* __restrict__ out, const T* __restrict__ in,
                                   long rows, int d) {
  constexpr int VE = Vec8<T>::kElems;
  const long total = rows * (d / VE);
  for (long idx = blockIdx.x * (long)blockDim.x + threadIdx.x; idx < total;
       idx += (long)gridDim.x * blockDim.x) {
    const long row = idx / (d / VE);
    const int col = (int)(idx % (d / VE)) * VE;
    Vec8<T> a = load16(in + row * 2 * d + col);
    Vec8<T> b = load16(in + row * 2 * d + d + col);
    Vec8<T> r;
#pragma unroll
    for (int j = 0; j < VE; ++j) {
      float g = to_f32(a.data[j]);
      float act;
      if (GELU) {
        // tanh-approx GELU (HF "gelu_pytorch_tanh")
        const float c = 0.7978845608028654f;  // sqrt(2/pi)
        float inner = c * (g + 0.044715f * g * g * g);
        act = 0.5f * g * (1.0f + tanhf(inner));
      } else {
        act = g / (1.0f + __expf(-g));  // SiLU
      }
      r.data[j] = from_f32<T>(act * to_f32(b.data[j]));
    }
    store16(out + row * d + col, r);
  }
}

// ----------------------------------------------------------------- RoPE ---
// NeoX-style rotate-half applied in place to q [T, Hq, D] and k [T, Hk, D].
// cos_sin: [max_pos, D] f32, first half cos(theta_i), second half sin.
// Grid: one workgroup per token; threads cover (head, dim-pair).

template <typename T>
__global__ void rope_kernel(T* __restrict__ q, T* __restrict__ k,
                            const long* __restrict__ positions,
                            const float* __restrict__ cos_sin,
                            int num_q_heads, int num_k_heads, int head_dim,
                            long q_stride, long k_stride) {
  const int token = blockIdx.x;
  const long pos = positions[token];
  const float* cs = cos_sin + pos * head_dim;
  const int half = head_dim / 2;
  const int total = (num_q_heads + num_k_heads) * half;
  for (int i = threadIdx.x; i < total; i += blockDim.x) {
    const int h = i / half;
    const int d = i % half;
    T* base = (h < num_q_heads)
                  ? q + (long)token * q_stride + (long)h * head_dim
                  : k + (long)token * k_stride + (long)(h - num_q_heads) * head_dim;
    const float c = cs[d];
    const float s = cs[half + d];
    const float x1 = to_f32(base[d]);
    const float x2 = to_f32(base[half + d]);
    base[d] = from_f32<T>(x1 * c - x2 * s);
    base[half + d] = from_f32<T>(x2 * c + x1 * s);
  }
}

// Gemma-2 "sandwich": x = rmsnorm(residual + rmsnorm(x, w_post), w_pre);
// residual += rmsnorm(x_in, w_post). One kernel replaces the post-block
// rmsnorm + fused_add_rmsnorm pair (two of these per layer — the decode
// step is launch-bound on these small norms at M=256).
template <typename T>
__global__ void norm_add_norm_kernel(T* __restrict__ x, T* __restrict__ residual,
                                     const T* __restrict__ w_post,
                                     const T* __restrict__ w_pre, int hidden,
                                     float eps, float offset) {
  constexpr int VE = Vec8<T>::kElems;
  __shared__ float red[4];
  const int row = blockIdx.x;
  T* xr = x + (long)row * hidden;
  T* rr = residual + (long)row * hidden;

  float ss = 0.f;
  for (int i = threadIdx.x * VE; i < hidden; i += blockDim.x * VE) {
    Vec8<T> v = load16(xr + i);
#pragma unroll
    for (int j = 0; j < VE; ++j) {
      float f = to_f32(v.data[j]);
      ss += f * f;
    }
  }
  ss = block_reduce_sum(ss, red);
  const float inv1 = rsqrtf(ss / hidden + eps);

  // residual += norm_post(x); accumulate the new residual's sumsq
  float ss2 = 0.f;
  for (int i = threadIdx.x * VE; i < hidden; i += blockDim.x * VE) {
    Vec8<T> vx = load16(xr + i);
    Vec8<T> vr = load16(rr + i);
    Vec8<T> vw = load16(w_post + i);
    Vec8<T> nr;
#pragma unroll
    for (int j = 0; j < VE; ++j) {
      const float t = to_f32(vx.data[j]) * inv1 * (to_f32(vw.data[j]) + offset);
      const float r = to_f32(vr.data[j]) + t;
      nr.data[j] = from_f32<T>(r);
      ss2 += r * r;
    }
    store16(rr + i, nr);
  }
  ss2 = block_reduce_sum(ss2, red);
  const float inv2 = rsqrtf(ss2 / hidden + eps);

  for (int i = threadIdx.x * VE; i < hidden; i += blockDim.x * VE) {
    Vec8<T> vr = load16(rr + i);
    Vec8<T> vw = load16(w_pre + i);
    Vec8<T> o;
#pragma unroll
    for (int j = 0; j < VE; ++j)
      o.data[j] = from_f32<T>(to_f32(vr.data[j]) * inv2 * (to_f32(vw.data[j]) + offset));
    store16(xr + i, o);
  }
}
