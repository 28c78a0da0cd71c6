// Paged-KV scatter: write this step's k/v vectors into the paged cache.
// Cache layout: [num_blocks, kv_heads, block_size, head_dim] (a (block,head)
// pair is a contiguous [block_size, head_dim] tile for the decode kernel).

#include "common.h"

template <typename T, typename TC = T>
__global__ void reshape_and_cache_kernel(
    const T* __restrict__ key,     // [T, kv_heads, D] (row stride key_stride)
    const T* __restrict__ value,
    TC* __restrict__ k_cache,      // [blocks, kv_heads, bs, D]
    TC* __restrict__ v_cache,
    const long* __restrict__ slots,  // [T] global slot id
    int kv_heads, int head_dim, int block_size,
    long key_stride, long val_stride) {
  constexpr int VE = Vec8<T>::kElems;
  const int token = blockIdx.x;
  const long slot = slots[token];
  if (slot < 0) return;  // padding
  const long block = slot / block_size;
  const int off = (int)(slot % block_size);
  const int chunks = (kv_heads * head_dim) / VE;
  for (int i = threadIdx.x; i < chunks; i += blockDim.x) {
    const int h = (i * VE) / head_dim;
    const int d = (i * VE) % head_dim;
    const long dst =
        ((block * kv_heads + h) * (long)block_size + off) * head_dim + d;
    Vec8<T> kv = load16(key + (long)token * key_stride + (long)h * head_dim + d);
    Vec8<T> vv = load16(value + (long)token * val_stride + (long)h * head_dim + d);
    if constexpr (std::is_same_v<T, TC>) {
      store16(k_cache + dst, kv);
      store16(v_cache + dst, vv);
    } else {  // quantizing cache write (e.g. bf16 → fp8)
#pragma unroll
      for (int j = 0; j < VE; ++j) {
        k_cache[dst + j] = from_f32<TC>(to_f32(kv.data[j]));
        v_cache[dst + j] = from_f32<TC>(to_f32(vv.data[j]));
      }
    }
  }
}

// Fused RoPE + cache write: rotates q and k in place AND scatters the
// rotated k plus v into the paged cache in one launch (the decode step is
// launch-bound on these per-layer elementwise kernels at M=256).
// Grid: one workgroup per token.
template <typename T, typename TC = T>
__global__ void rope_and_cache_kernel(
    T* __restrict__ q,             // [T, Hq, D] (row stride q_stride)
    T* __restrict__ k,             // [T, Hk, D]
    const T* __restrict__ value,   // [T, Hk, D]
    TC* __restrict__ k_cache,      // [blocks, Hk, bs, D]
    TC* __restrict__ v_cache,
    const long* __restrict__ positions,
    const float* __restrict__ cos_sin,  // [max_pos, D] f32
    const long* __restrict__ slots,
    int num_q_heads, int num_k_heads, int head_dim, int block_size,
    long q_stride, long k_stride, long v_stride) {
  const int token = blockIdx.x;
  const long pos = positions[token];
  const float* cs = cos_sin + pos * head_dim;
  const int half = head_dim / 2;
  const long slot = slots[token];
  const long block = slot / block_size;
  const int off = (int)(slot % block_size);

  // rotate q heads; rotate k heads and write both halves to the cache
  const int total = (num_q_heads + num_k_heads) * half;
  for (int i = threadIdx.x; i < total; i += blockDim.x) {
    const int h = i / half;
    const int d = i % half;
    const float c = cs[d];
    const float s = cs[half + d];
    if (h < num_q_heads) {
      T* base = q + (long)token * q_stride + (long)h * head_dim;
      const float x1 = to_f32(base[d]);
      const float x2 = to_f32(base[half + d]);
      base[d] = from_f32<T>(x1 * c - x2 * s);
      base[half + d] = from_f32<T>(x2 * c + x1 * s);
    } else {
      const int kh = h - num_q_heads;
      T* base = k + (long)token * k_stride + (long)kh * head_dim;
      const float x1 = to_f32(base[d]);
      const float x2 = to_f32(base[half + d]);
      const T r1 = from_f32<T>(x1 * c - x2 * s);
      const T r2 = from_f32<T>(x2 * c + x1 * s);
      base[d] = r1;
      base[half + d] = r2;
      if (slot >= 0) {
        TC* kdst = k_cache +
            ((block * num_k_heads + kh) * (long)block_size + off) * head_dim;
        if constexpr (std::is_same_v<T, TC>) {
          kdst[d] = r1;
          kdst[half + d] = r2;
        } else {
          kdst[d] = from_f32<TC>(to_f32(r1));
          kdst[half + d] = from_f32<TC>(to_f32(r2));
        }
      }
    }
  }
  if (slot < 0) return;
  // copy v (vectorized; untouched by rope)
  constexpr int VE = Vec8<T>::kElems;
  const int vchunks = (num_k_heads * head_dim) / VE;
  for (int i = threadIdx.x; i < vchunks; i += blockDim.x) {
    const int h = (i * VE) / head_dim;
    const int d = (i * VE) % head_dim;
    Vec8<T> vv = load16(value + (long)token * v_stride + (long)h * head_dim + d);
    TC* vdst = v_cache +
        ((block * num_k_heads + h) * (long)block_size + off) * head_dim + d;
    if constexpr (std::is_same_v<T, TC>) {
      store16(vdst, vv);
    } else {
#pragma unroll
      for (int j = 0; j < VE; ++j) vdst[j] = from_f32<TC>(to_f32(vv.data[j]));
    }
  }
}
