// Paged attention for gfx950.
//
// DECODE (paged_decode_attention_kernel): the serving hot path. One
// workgroup per (sequence, kv_head); all GQA query heads of that kv head
// are processed together so the KV stream is read ONCE (decode attention
// at D=128/G<=8 is ~4 FLOP/byte — far below the 25:1 VALU roofline, so the
// kernel is designed as a clean HBM stream: 16-byte coalesced K loads,
// two-phase chunk processing with scores staged in LDS, fp32 online
// softmax).
//
//   phase A: 16-lane groups compute q·k for CHUNK tokens -> s[CHUNK][G] LDS
//   phase B: thread (h, d-slice) runs online softmax over the chunk and
//            accumulates p·V into per-thread fp32 registers; V rows are
//            read coalesced by each 32-thread head-group (L1 serves the
//            G-way reuse across head-groups).
//
// PREFILL (varlen_prefill_attention_kernel): correctness-first packed
// varlen causal attention (one wave per query row, keys across lanes,
// online softmax). GEMM-shaped MFMA flash prefill is the planned
// replacement; at the serving shapes prefill attention is a small share of
// prefill FLOPs (see SURVEY §7 risk list).

#include "common.h"

#define DECODE_CHUNK 64  // tokens per two-phase iteration (4 KV blocks @ bs 16)

template <typename T, int HEAD_DIM>
__global__ __launch_bounds__(256) void paged_decode_attention_kernel(
    T* __restrict__ out,                  // [B, H, D] (row stride out_stride)
    const T* __restrict__ q,              // [B, H, D] (row stride q_stride)
    const T* __restrict__ k_cache,        // [nb, KVH, bs, D]
    const T* __restrict__ v_cache,
    const int* __restrict__ block_tables, // [B, max_blocks]
    const int* __restrict__ context_lens, // [B]
    int num_heads, int num_kv_heads, int block_size, int max_blocks,
    float scale, float softcap, int window,
    long q_stride, long out_stride) {
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int G = num_heads / num_kv_heads;  // <= 8
  const int L = context_lens[b];
  if (L <= 0) return;

  constexpr int VE = Vec8<T>::kElems;        // 8 for bf16, 4 for f32
  constexpr int D = HEAD_DIM;
  const int lane16 = threadIdx.x & 15;       // lane within 16-lane score group
  const int g16 = threadIdx.x >> 4;          // score group id (0..15)

  // LDS: q (G x D f32, pre-scaled), scores (CHUNK x G f32)
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* q_s = reinterpret_cast<float*>(smem);              // [G][D]
  float* s_s = q_s + G * D;                                 // [CHUNK][G]

  // Stage q, pre-scaled by `scale`.
  for (int i = threadIdx.x; i < G * D; i += blockDim.x) {
    const int h = i / D, d = i % D;
    q_s[i] = to_f32(q[(long)b * q_stride + (long)(kh * G + h) * D + d]) * scale;
  }
  __syncthreads();

  const int start = (window > 0 && L > window) ? (L - window) : 0;
  const int* bt = block_tables + (long)b * max_blocks;

  // Phase-B per-thread state: head h, dims [d0, d0 + DPT)
  constexpr int DPT = D / 32;               // dims per thread (2/4/8)
  const int h_b = threadIdx.x >> 5;         // 0..7
  const int d0 = (threadIdx.x & 31) * DPT;
  float acc[DPT];
#pragma unroll
  for (int i = 0; i < DPT; ++i) acc[i] = 0.f;
  float m_run = -1e30f, l_run = 0.f;

  for (int base = (start / DECODE_CHUNK) * DECODE_CHUNK; base < L;
       base += DECODE_CHUNK) {
    const int chunk_end = min(base + DECODE_CHUNK, L);
    // ---- phase A: scores for tokens [base, chunk_end)
    for (int t = base + g16; t < chunk_end; t += 16) {
      const long blk = bt[t / block_size];
      const int off = t % block_size;
      const T* krow =
          k_cache + (((blk * num_kv_heads + kh) * (long)block_size + off)) * D;
      // lane16 covers dims [lane16*D/16, ...): D/16 elems = 16B at D=128/bf16
      constexpr int DL = D / 16;
      float kf[DL];
      if constexpr (DL % VE == 0) {
#pragma unroll
        for (int c = 0; c < DL; c += VE) {
          Vec8<T> kv = load16(krow + lane16 * DL + c);
#pragma unroll
          for (int j = 0; j < VE; ++j) kf[c + j] = to_f32(kv.data[j]);
        }
      } else {  // e.g. D=64 bf16: 4 elems per lane — scalar loads
#pragma unroll
        for (int c = 0; c < DL; ++c) kf[c] = to_f32(krow[lane16 * DL + c]);
      }
      for (int h = 0; h < G; ++h) {
        float dot = 0.f;
        const float* qrow = q_s + h * D + lane16 * DL;
#pragma unroll
        for (int j = 0; j < DL; ++j) dot += qrow[j] * kf[j];
        dot = group_reduce_sum<16>(dot);
        if (lane16 == 0) {
          if (softcap > 0.f) dot = tanhf(dot / softcap) * softcap;
          if (t < start) dot = -1e30f;
          s_s[(t - base) * G + h] = dot;
        }
      }
    }
    __syncthreads();
    // ---- phase B: online softmax + V accumulation
    if (h_b < G) {
      const int n = chunk_end - base;
      float m_chunk = -1e30f;
      for (int t = 0; t < n; ++t) m_chunk = fmaxf(m_chunk, s_s[t * G + h_b]);
      const float m_new = fmaxf(m_run, m_chunk);
      if (m_new > -1e30f) {
        const float alpha = __expf(m_run - m_new);
#pragma unroll
        for (int i = 0; i < DPT; ++i) acc[i] *= alpha;
        l_run *= alpha;
        m_run = m_new;
        for (int t = 0; t < n; ++t) {
          const float p = __expf(s_s[t * G + h_b] - m_new);
          l_run += p;
          const int tok = base + t;
          const long blk = bt[tok / block_size];
          const int off = tok % block_size;
          const T* vrow =
              v_cache + (((blk * num_kv_heads + kh) * (long)block_size + off)) * D + d0;
#pragma unroll
          for (int i = 0; i < DPT; ++i) acc[i] += p * to_f32(vrow[i]);
        }
      }
    }
    __syncthreads();  // protect s_s for the next chunk
  }

  if (h_b < G) {
    const float inv = (l_run > 0.f) ? 1.0f / l_run : 0.f;
    T* orow = out + (long)b * out_stride + (long)(kh * G + h_b) * D + d0;
#pragma unroll
    for (int i = 0; i < DPT; ++i) orow[i] = from_f32<T>(acc[i] * inv);
  }
}

// --------------------------------------------------------------- prefill --
// Packed varlen causal attention. Grid: (seq, head). 4 waves per block;
// each wave owns query rows i = wave_id + 4*n. Keys go across lanes (one
// key per lane per 64-key chunk), V accumulation is lane-dim-parallel via
// shfl broadcast of the probabilities.

template <typename T, int HEAD_DIM>
__global__ __launch_bounds__(256) void varlen_prefill_attention_kernel(
    T* __restrict__ out,            // [T, H, D]
    const T* __restrict__ q,        // [T, H, D] (row strides below)
    const T* __restrict__ k,        // [T, KVH, D]
    const T* __restrict__ v,
    const int* __restrict__ cu_seqlens,  // [B+1]
    int num_heads, int num_kv_heads, float scale, float softcap, int window,
    long q_stride, long k_stride, long v_stride, long o_stride) {
  constexpr int D = HEAD_DIM;
  constexpr int VE = Vec8<T>::kElems;
  constexpr int DPT = D / WAVE;  // dims per lane (1/2/4)
  const int seq = blockIdx.x;
  const int h = blockIdx.y;
  const int kvh = h / (num_heads / num_kv_heads);
  const int s0 = cu_seqlens[seq], s1 = cu_seqlens[seq + 1];
  const int L = s1 - s0;
  const int wid = threadIdx.x / WAVE;
  const int lane = threadIdx.x & (WAVE - 1);

  __shared__ __attribute__((aligned(16))) float q_lds[4][D];

  for (int i = wid; i < L; i += 4) {  // query row
    // stage q row (pre-scaled) into this wave's LDS slot
    for (int d = lane; d < D; d += WAVE)
      q_lds[wid][d] = to_f32(q[(long)(s0 + i) * q_stride + (long)h * D + d]) * scale;
    __builtin_amdgcn_wave_barrier();

    float m_run = -1e30f, l_run = 0.f;
    float acc[DPT];
#pragma unroll
    for (int t = 0; t < DPT; ++t) acc[t] = 0.f;

    const int kstart = (window > 0 && i + 1 > window) ? (i + 1 - window) : 0;
    for (int base = (kstart / WAVE) * WAVE; base <= i; base += WAVE) {
      const int j = base + lane;  // this lane's key
      float score = -1e30f;
      if (j <= i && j >= kstart && j < L) {
        const T* krow = k + (long)(s0 + j) * k_stride + (long)kvh * D;
        float dot = 0.f;
#pragma unroll 4
        for (int d = 0; d < D; d += VE) {
          Vec8<T> kv = load16(krow + d);
#pragma unroll
          for (int e = 0; e < VE; ++e) dot += q_lds[wid][d + e] * to_f32(kv.data[e]);
        }
        score = (softcap > 0.f) ? tanhf(dot / softcap) * softcap : dot;
      }
      const float m_chunk = wave_reduce_max(score);
      const float m_new = fmaxf(m_run, m_chunk);
      const float alpha = __expf(m_run - m_new);
      const float p = (score > -1e29f) ? __expf(score - m_new) : 0.f;
      const float psum = wave_reduce_sum(p);
      l_run = l_run * alpha + psum;
      m_run = m_new;
#pragma unroll
      for (int t = 0; t < DPT; ++t) acc[t] *= alpha;
      const int jn = min(i - base + 1, WAVE);
      for (int jj = 0; jj < jn; ++jj) {
        const float pj = __shfl(p, jj, WAVE);
        if (pj != 0.f) {
          const T* vrow = v + (long)(s0 + base + jj) * v_stride + (long)kvh * D;
#pragma unroll
          for (int t = 0; t < DPT; ++t)
            acc[t] += pj * to_f32(vrow[lane * DPT + t]);
        }
      }
    }
    const float inv = (l_run > 0.f) ? 1.0f / l_run : 0.f;
    T* orow = out + (long)(s0 + i) * o_stride + (long)h * D;
#pragma unroll
    for (int t = 0; t < DPT; ++t) orow[lane * DPT + t] = from_f32<T>(acc[t] * inv);
    __builtin_amdgcn_wave_barrier();
  }
}
