// Paged-KV scatter: write this step's k/v vectors into the paged cache.
// Cache layout: [num_blocks, kv_heads, block_size, head_dim] (a (block,head)
// pair is a contiguous [block_size, head_dim] tile for the decode kernel).

#include "common.h"

template <typename T>
__global__ void reshape_and_cache_kernel(
    const T* __restrict__ key,     // [T, kv_heads, D] (row stride key_stride)
    const T* __restrict__ value,
    T* __restrict__ k_cache,       // [blocks, kv_heads, bs, D]
    T* __restrict__ v_cache,
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
    store16(k_cache + dst, kv);
    Vec8<T> vv = load16(value + (long)token * val_stride + (long)h * head_dim + d);
    store16(v_cache + dst, vv);
  }
}
