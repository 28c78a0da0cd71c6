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

#define DECODE_CHUNK 128  // tokens per two-phase iteration (8 KV blocks @ bs 16)

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
  constexpr int CHUNK = DECODE_CHUNK;
  const int lane16 = threadIdx.x & 15;       // lane within 16-lane score group
  const int g16 = threadIdx.x >> 4;          // score group id (0..15)

  // Phase-B decomposition: SUBS token-parallel sub-groups per head, each of
  // 32 threads covering all D dims. All 256 threads work for G*32*SUBS==256.
  const int tph = 32 * G;                    // threads per head-set
  const int SUBS = 256 / tph;                // 8,4,2,2,1,1,1,1 for G=1..8
  const int sub = threadIdx.x / tph;         // token sub-group (>=SUBS: idle)
  const int h_b = (threadIdx.x % tph) >> 5;  // head
  constexpr int DPT = D / 32;                // dims per thread (2/4/8)
  const int d0 = (threadIdx.x & 31) * DPT;

  // LDS: q (G x D f32, pre-scaled) | scores (CHUNK x G) | chunk block ids |
  //      merge buffers (SUBS x G x (D + 2))
  extern __shared__ __attribute__((aligned(16))) char smem[];
  float* q_s = reinterpret_cast<float*>(smem);              // [G][D]
  float* s_s = q_s + G * D;                                 // [CHUNK][G]
  int* bt_s = reinterpret_cast<int*>(s_s + CHUNK * G);      // [CHUNK/bs up to 16]
  float* merge_s = reinterpret_cast<float*>(bt_s + 16);     // [SUBS][G][D+2]

  for (int i = threadIdx.x; i < G * D; i += blockDim.x) {
    const int h = i / D, d = i % D;
    q_s[i] = to_f32(q[(long)b * q_stride + (long)(kh * G + h) * D + d]) * scale;
  }

  const int start = (window > 0 && L > window) ? (L - window) : 0;
  const int* bt = block_tables + (long)b * max_blocks;
  const int base0 = (start / CHUNK) * CHUNK;

  float acc[DPT];
#pragma unroll
  for (int i = 0; i < DPT; ++i) acc[i] = 0.f;
  float m_run = -1e30f, l_run = 0.f;

  for (int base = base0; base < L; base += CHUNK) {
    const int chunk_end = min(base + CHUNK, L);
    const int n = chunk_end - base;
    // stage this chunk's block ids (removes a dependent global load from the
    // phase-B address chain)
    if (threadIdx.x < (unsigned)((n + block_size - 1) / block_size))
      bt_s[threadIdx.x] = bt[(base + threadIdx.x * block_size) / block_size];
    __syncthreads();
    // ---- phase A: scores for tokens [base, chunk_end)
    for (int t0 = g16; t0 < n; t0 += 16) {
      const int t = base + t0;
      const long blk = bt_s[t0 / block_size];
      const int off = t % block_size;
      const T* krow =
          k_cache + (((blk * num_kv_heads + kh) * (long)block_size + off)) * D;
      constexpr int DL = D / 16;  // elems per lane16
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
          s_s[t0 * G + h] = dot;
        }
      }
    }
    __syncthreads();
    // ---- phase B: online softmax + V accumulation (sub-group token split)
    if (sub < SUBS) {
      float m_chunk = -1e30f;
      for (int t = sub; t < n; t += SUBS)
        m_chunk = fmaxf(m_chunk, s_s[t * G + h_b]);
      const float m_new = fmaxf(m_run, m_chunk);
      if (m_new > -1e30f) {
        const float alpha = __expf(m_run - m_new);
#pragma unroll
        for (int i = 0; i < DPT; ++i) acc[i] *= alpha;
        l_run *= alpha;
        m_run = m_new;
        // Iterations are independent (chunk-max softmax, no rescale inside)
        // so V loads pipeline across tokens.
        for (int t = sub; t < n; t += SUBS) {
          const float p = __expf(s_s[t * G + h_b] - m_new);
          l_run += p;
          const int tok = base + t;
          const long blk = bt_s[t / block_size];
          const int off = tok % block_size;
          const T* vrow =
              v_cache + (((blk * num_kv_heads + kh) * (long)block_size + off)) * D + d0;
          // Vectorized V row slice: DPT contiguous elems per thread
          // (16B for bf16 D=256, 8B for D=128) — scalar 2B loads here
          // were the v2 bandwidth ceiling.
          if constexpr (DPT % Vec8<T>::kElems == 0) {
            constexpr int VE8 = Vec8<T>::kElems;
#pragma unroll
            for (int c = 0; c < DPT; c += VE8) {
              Vec8<T> vv = load16(vrow + c);
#pragma unroll
              for (int j = 0; j < VE8; ++j) acc[c + j] += p * to_f32(vv.data[j]);
            }
          } else if constexpr (DPT % Vec4<T>::kElems == 0) {
            constexpr int VE4 = Vec4<T>::kElems;
#pragma unroll
            for (int c = 0; c < DPT; c += VE4) {
              Vec4<T> vv = load8(vrow + c);
#pragma unroll
              for (int j = 0; j < VE4; ++j) acc[c + j] += p * to_f32(vv.data[j]);
            }
          } else {
#pragma unroll
            for (int i = 0; i < DPT; ++i) acc[i] += p * to_f32(vrow[i]);
          }
        }
      }
    }
    __syncthreads();  // protect s_s / bt_s for the next chunk
  }

  // ---- merge sub-group partials (flash-decoding style, within the block)
  const int d2 = D + 2;
  if (sub < SUBS && SUBS > 1) {
    float* slot = merge_s + (sub * G + h_b) * d2;
#pragma unroll
    for (int i = 0; i < DPT; ++i) slot[d0 + i] = acc[i];
    if (d0 == 0) {
      slot[D] = m_run;
      slot[D + 1] = l_run;
    }
  }
  __syncthreads();
  if (sub == 0) {
    float m_tot = m_run, l_tot = 0.f;
    if (SUBS > 1) {
      for (int s2 = 0; s2 < SUBS; ++s2)
        m_tot = fmaxf(m_tot, merge_s[(s2 * G + h_b) * d2 + D]);
      float accm[DPT];
#pragma unroll
      for (int i = 0; i < DPT; ++i) accm[i] = 0.f;
      for (int s2 = 0; s2 < SUBS; ++s2) {
        const float* slot = merge_s + (s2 * G + h_b) * d2;
        const float w = __expf(slot[D] - m_tot);
        l_tot += w * slot[D + 1];
#pragma unroll
        for (int i = 0; i < DPT; ++i) accm[i] += w * slot[d0 + i];
      }
#pragma unroll
      for (int i = 0; i < DPT; ++i) acc[i] = accm[i];
    } else {
      l_tot = l_run;
    }
    const float inv = (l_tot > 0.f) ? 1.0f / l_tot : 0.f;
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
    T* __restrict__ out,            // [Tq, H, D]
    const T* __restrict__ q,        // [Tq, H, D] (row strides below)
    const T* __restrict__ k,        // [Tk, KVH, D]
    const T* __restrict__ v,
    const int* __restrict__ cu_seqlens,    // [B+1] query offsets
    const int* __restrict__ cu_seqlens_k,  // [B+1] key offsets (chunked)
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
  const int s0k = cu_seqlens_k[seq];
  const int Lk = cu_seqlens_k[seq + 1] - s0k;
  const int off = Lk - L;  // abs position of q row 0
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

    const int iabs = off + i;  // absolute position of this query row
    const int kstart = (window > 0 && iabs + 1 > window) ? (iabs + 1 - window) : 0;
    for (int base = (kstart / WAVE) * WAVE; base <= iabs; base += WAVE) {
      const int j = base + lane;  // this lane's key (absolute)
      float score = -1e30f;
      if (j <= iabs && j >= kstart && j < Lk) {
        const T* krow = k + (long)(s0k + j) * k_stride + (long)kvh * D;
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
      const int jn = min(iabs - base + 1, WAVE);
      for (int jj = 0; jj < jn; ++jj) {
        const float pj = __shfl(p, jj, WAVE);
        if (pj != 0.f) {
          const T* vrow = v + (long)(s0k + base + jj) * v_stride + (long)kvh * D;
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

// ------------------------------------------------------ MFMA flash prefill --
// Packed varlen causal GQA attention on matrix cores (bf16).
// Grid: (seq, q_head, q_tile). Workgroup: 4 waves; each wave owns 16 query
// rows of a 64-row Q tile. KV tiles of 64 keys staged in XOR-swizzled LDS
// (shared by the 4 waves); per KV tile each wave computes
//   S[16q,64k]  via mfma_f32_16x16x32_bf16 (Q frags in registers, K^T frags
//               from LDS b128 reads)
//   online softmax in the MFMA C-layout (row r = (lane>>4)*4+reg lives on
//               the 16 col-lanes -> shfl_xor over the low 4 lane bits)
//   OT[D,16q] += mfma(A=V^T, B=P^T)  -- transposed PV so the P fragment is a
//               contiguous b128 read of the per-wave P_lds tile and the
//               softmax rescale factor is lane-uniform (qrow = lane&15).
// Epilogue transposes OT back with scalar stores.

typedef __attribute__((ext_vector_type(8))) short bf16x8_t;
typedef __attribute__((ext_vector_type(4))) float f32x4_t;
typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4v;
typedef __attribute__((address_space(3))) bf16x4v lds_bf16x4;

#define PF_QT 64   // q rows per workgroup
#define PF_KT 64   // keys per kv tile

DEVINL int swz(int row, int d) {  // element-index XOR swizzle (16B granules)
  return d ^ ((row & 7) << 3);
}

// cu_seqlens_k: key-side offsets — equal to cu_seqlens for plain prefill;
// for CHUNKED prefill a sequence's K/V covers its full context so far
// (gathered past + fresh chunk) while Q is only the new chunk, and q row i
// sits at absolute position (Lk - Lq) + i.
template <int HEAD_DIM>
__global__ __launch_bounds__(256) void flash_prefill_bf16_kernel(
    __hip_bfloat16* __restrict__ out,      // [Tq, H, D]
    const __hip_bfloat16* __restrict__ q,  // [Tq, H, D]
    const __hip_bfloat16* __restrict__ k,  // [Tk, KVH, D]
    const __hip_bfloat16* __restrict__ v,
    const int* __restrict__ cu_seqlens, const int* __restrict__ cu_seqlens_k,
    int num_heads, int num_kv_heads,
    float scale, float softcap, int window, long q_stride, long k_stride,
    long v_stride, long o_stride) {
  constexpr int D = HEAD_DIM;
  const int seq = blockIdx.x;
  const int h = blockIdx.y;
  const int kvh = h / (num_heads / num_kv_heads);
  const int s0 = cu_seqlens[seq];
  const int L = cu_seqlens[seq + 1] - s0;       // query rows
  const int s0k = cu_seqlens_k[seq];
  const int Lk = cu_seqlens_k[seq + 1] - s0k;   // key rows (>= L)
  const int off = Lk - L;                       // abs position of q row 0
  const int q_base = blockIdx.z * PF_QT;
  if (q_base >= L) return;

  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int col = lane & 15;        // MFMA col lane (key for S, qrow for OT)
  const int kgrp = lane >> 4;       // 0..3

  // LDS: K tile | V tile | per-wave P tiles | per-wave row stats
  __shared__ __attribute__((aligned(16))) short k_lds[PF_KT * D];
  // V: tr-subtile image, 528-elem stride per subtile (bank rotation)
  __shared__ __attribute__((aligned(16))) short v_lds[(PF_KT / 32) * (D / 16) * 528];
  __shared__ __attribute__((aligned(16))) short p_lds[4][16 * PF_KT];
  __shared__ float stat_lds[4][2][16];  // [wave][alpha|inv_l][row]

  // ---- load Q fragments (registers): frag[ks] covers dims ks*32+(kgrp*8..+7)
  constexpr int KS = D / 32;
  bf16x8_t qfrag[KS];
  const int my_qrow = q_base + wid * 16 + col;  // A-frag row = lane&15
  {
    const int r = (my_qrow < L) ? my_qrow : (L - 1);
    const __hip_bfloat16* qrow_p = q + (long)(s0 + r) * q_stride + (long)h * D;
#pragma unroll
    for (int ks = 0; ks < KS; ++ks)
      qfrag[ks] = *reinterpret_cast<const bf16x8_t*>(qrow_p + ks * 32 + kgrp * 8);
  }

  // per-lane softmax state for rows rr = wid*16 + kgrp*4 + reg
  float m_run[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  float l_run[4] = {0.f, 0.f, 0.f, 0.f};
  constexpr int DT = D / 16;
  f32x4_t ot[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) ot[dt] = {0.f, 0.f, 0.f, 0.f};

  const int wave_max_row = off + min(q_base + wid * 16 + 15, L - 1);
  const int block_max_row = off + min(q_base + PF_QT - 1, L - 1);
  const int kv_end = block_max_row + 1;  // causal bound (absolute keys)
  int kv_begin = 0;
  if (window > 0) {
    const int wave_min_needed = off + q_base + 1 - window;
    kv_begin = max(0, (wave_min_needed / PF_KT) * PF_KT);
  }

  for (int kt = kv_begin; kt < kv_end; kt += PF_KT) {
    // ---- stage K/V tile (zero-fill beyond L so garbage never reaches MFMA)
    {
      constexpr int CHW = 8;  // elems per 16B chunk
      const int chunks = PF_KT * D / CHW;
      for (int c = tid; c < chunks; c += 256) {
        const int key = c / (D / CHW);
        const int d8 = (c % (D / CHW)) * CHW;
        const int dst = key * D + swz(key, d8);
        // V uses the tr-read subtile image (quad-permuted 32-key × 16-dim
        // blocks) so PV A-fragments load via ds_read_b64_tr_b16
        const int dtile = d8 / 16, col0 = d8 & 15;
        const int qq = (key & 31) >> 2;
        const int bpos = ((qq & 1) << 2) + (qq >> 1);
        const int vdst = ((key >> 5) * (D / 16) + dtile) * 528 + bpos * 64 +
                         (key & 3) * 16 + col0;
        const int gkey = kt + key;
        if (gkey < Lk) {
          *reinterpret_cast<bf16x8_t*>(&k_lds[dst]) =
              *reinterpret_cast<const bf16x8_t*>(
                  k + (long)(s0k + gkey) * k_stride + (long)kvh * D + d8);
          *reinterpret_cast<bf16x8_t*>(&v_lds[vdst]) =
              *reinterpret_cast<const bf16x8_t*>(
                  v + (long)(s0k + gkey) * v_stride + (long)kvh * D + d8);
        } else {
          *reinterpret_cast<bf16x8_t*>(&k_lds[dst]) = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
          *reinterpret_cast<bf16x8_t*>(&v_lds[vdst]) = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
        }
      }
    }
    __syncthreads();

    if (kt <= wave_max_row) {  // this wave has rows that see this tile
      // ---- S = Q K^T for 4 column tiles of 16 keys
      f32x4_t s[4];
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        s[ct] = {0.f, 0.f, 0.f, 0.f};
        const int key = ct * 16 + col;
#pragma unroll
        for (int ks = 0; ks < KS; ++ks) {
          const int d8 = ks * 32 + kgrp * 8;
          bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
              &k_lds[key * D + swz(key, d8)]);
          s[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[ks], bfrag, s[ct], 0, 0, 0);
        }
      }
      // ---- scale, softcap, mask; rows rr = wid*16+kgrp*4+reg
      float m_tile[4], l_tile[4], p[4][4];
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int qrow = q_base + wid * 16 + kgrp * 4 + reg;
        const int row = off + qrow;  // absolute position
        float mx = -1e30f;
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
          const int key = kt + ct * 16 + col;
          float x = s[ct][reg] * scale;
          if (softcap > 0.f) x = tanhf(x / softcap) * softcap;
          const bool dead = key > row || key >= Lk || qrow >= L ||
                            (window > 0 && key <= row - window);
          x = dead ? -1e30f : x;
          p[reg][ct] = x;
          mx = fmaxf(mx, x);
        }
        // reduce max over the 16 col lanes
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) mx = fmaxf(mx, __shfl_xor(mx, off, WAVE));
        m_tile[reg] = mx;
      }
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const float m_new = fmaxf(m_run[reg], m_tile[reg]);
        const float alpha = (m_new > -1e30f) ? __expf(m_run[reg] - m_new) : 1.f;
        float lsum = 0.f;
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
          const float pe = (m_new > -1e30f && p[reg][ct] > -1e29f)
                               ? __expf(p[reg][ct] - m_new) : 0.f;
          p[reg][ct] = pe;
          lsum += pe;
        }
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) lsum += __shfl_xor(lsum, off, WAVE);
        l_run[reg] = l_run[reg] * alpha + lsum;
        m_run[reg] = m_new;
        // stash alpha for the OT lanes (indexed by qrow = lane&15)
        if (col == 0) stat_lds[wid][0][kgrp * 4 + reg] = alpha;
        // write P row to the wave's P tile (bf16)
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
          p_lds[wid][(kgrp * 4 + reg) * PF_KT + ct * 16 + col] =
              __bfloat16_as_short(__float2bfloat16(p[reg][ct]));
        }
      }
      // ---- OT += V^T P^T  (A = V^T frags via scalar LDS reads, B = P^T b128)
      const float alpha_q = stat_lds[wid][0][col];
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        ot[dt][0] *= alpha_q; ot[dt][1] *= alpha_q;
        ot[dt][2] *= alpha_q; ot[dt][3] *= alpha_q;
#pragma unroll
        for (int ks = 0; ks < PF_KT / 32; ++ks) {
          const int sub = (ks * (D / 16) + dt) * 528 + lane * 4;
          bf16x4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (lds_bf16x4*)&v_lds[sub]);
          bf16x4v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (lds_bf16x4*)&v_lds[sub + 4 * 64]);
          bf16x8_t a;
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            a[j] = __bfloat16_as_short((__hip_bfloat16)lo[j]);
            a[4 + j] = __bfloat16_as_short((__hip_bfloat16)hi[j]);
          }
          bf16x8_t b = *reinterpret_cast<const bf16x8_t*>(
              &p_lds[wid][col * PF_KT + ks * 32 + kgrp * 8]);
          ot[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, ot[dt], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: normalise and store (transpose OT back per lane)
  if (col == 0) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const float l = l_run[reg];
      stat_lds[wid][1][kgrp * 4 + reg] = (l > 0.f) ? 1.0f / l : 0.f;
    }
  }
  __builtin_amdgcn_s_waitcnt(0);  // lgkm drain before same-wave read
  const float inv = stat_lds[wid][1][col];
  const int orow = q_base + wid * 16 + col;
  if (orow < L) {
    __hip_bfloat16* op = out + (long)(s0 + orow) * o_stride + (long)h * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        op[dt * 16 + kgrp * 4 + reg] = __float2bfloat16(ot[dt][reg] * inv);
    }
  }
}

// 16-B-per-lane global→LDS DMA, HIDDEN from hipcc in inline asm: with the
// builtin form the compiler conservatively re-inserts `s_waitcnt vmcnt(0)`
// before the first ds_read that may alias the DMA destination (verified in
// the .s: it landed ahead of the PV tr16 reads, draining the K prefetch
// every chunk). Hand-counted vmcnt in pipe_barrier_vm is the only wait
// discipline for these, per guide §5.7 (its LDS-DMA recipe: M0 carries the
// wave-uniform LDS byte address, saved/written/restored in ONE statement;
// the s_nop is the SALU-write→M0-read wait state).
DEVINL void glds16(const void* src, void* lds_dst_uniform) {
  unsigned lds_off = (unsigned)(unsigned long)(
      (__attribute__((address_space(3))) char*)lds_dst_uniform);
  lds_off = __builtin_amdgcn_readfirstlane(lds_off);
  unsigned keep;
  asm volatile(
      "s_mov_b32 %0, m0\n\t"
      "s_mov_b32 m0, %2\n\t"
      "s_nop 0\n\t"
      "global_load_lds_dwordx4 %1, off\n\t"
      "s_mov_b32 m0, %0"
      : "=&s"(keep)
      : "v"(src), "s"(lds_off)
      : "memory");
}

DEVINL void pipe_barrier() {  // lgkm drain + raw barrier (glds may span)
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
}

template <int N>
DEVINL void pipe_barrier_vm() {  // + counted vmcnt: N glds may stay in flight
  asm volatile("s_waitcnt vmcnt(%0) lgkmcnt(0)" :: "i"(N) : "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
}

DEVINL void reg_fence(bf16x8_t& v) {  // force materialisation (wait) HERE
  asm volatile("" : "+v"(v));
}


// ---------------------------------------------- pipelined MFMA flash prefill --
// flash_prefill_bf16_kernel with the decode kernel's glds software pipeline
// (see paged_decode_pipe_kernel below for the full rationale): the serial
// stage→sync→S→softmax→PV structure exposes the K/V staging latency every
// tile (L2-resident for the reuse across q-heads/q-tiles, but the VGPR
// round-trip + barrier drain still serialize). Two buffers: V(kt)
// overwrites K(kt)'s buffer after S, K(kt+1) prefetches into the other;
// raw barriers with counted vmcnt keep the DMA in flight through PV.
// The tile containing Lk falls back to plain zero-filled staging.

template <int HEAD_DIM>
__global__ __launch_bounds__(256, (HEAD_DIM <= 128 ? 4 : 2)) void flash_prefill_pipe_kernel(
    __hip_bfloat16* __restrict__ out,      // [Tq, H, D]
    const __hip_bfloat16* __restrict__ q,  // [Tq, H, D]
    const __hip_bfloat16* __restrict__ k,  // [Tk, KVH, D]
    const __hip_bfloat16* __restrict__ v,
    const int* __restrict__ cu_seqlens, const int* __restrict__ cu_seqlens_k,
    int num_heads, int num_kv_heads,
    float scale, float softcap, int window, long q_stride, long k_stride,
    long v_stride, long o_stride) {
  constexpr int D = HEAD_DIM;
  constexpr int NT = 256;
  constexpr int CPK = D / 8;                        // 16B granules per key row
  constexpr int NI_K = PF_KT * CPK / NT;            // K glds per wave per tile
  constexpr int NSUB = (PF_KT / 32) * (D / 16);     // V tr-subtiles per tile
  constexpr int NI_V = NSUB / 4;                    // V glds per wave per tile
  constexpr int KV_ELEMS = (PF_KT * D > NSUB * 528) ? PF_KT * D : NSUB * 528;
  const int seq = blockIdx.x;
  const int h = blockIdx.y;
  const int kvh = h / (num_heads / num_kv_heads);
  const int s0 = cu_seqlens[seq];
  const int L = cu_seqlens[seq + 1] - s0;       // query rows
  const int s0k = cu_seqlens_k[seq];
  const int Lk = cu_seqlens_k[seq + 1] - s0k;   // key rows (>= L)
  const int off = Lk - L;                       // abs position of q row 0
  const int q_base = blockIdx.z * PF_QT;
  if (q_base >= L) return;

  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int col = lane & 15;
  const int kgrp = lane >> 4;

  // ONE shared array (glds pipeline: hipcc drains vmcnt(0) before ds_reads
  // when a second __shared__ object exists)
  constexpr int P_OFF = 2 * KV_ELEMS;                    // shorts
  // float index of the stat region: 16B-align past P (8 shorts per 16B
  // unit, 4 floats per unit)
  constexpr int STAT_F = (P_OFF + 4 * 16 * PF_KT + 7) / 8 * 4;
  constexpr int TOTAL_BYTES = STAT_F * 4 + 4 * 2 * 16 * 4;
  __shared__ __attribute__((aligned(16))) char smem[TOTAL_BYTES];
  short* const kv0 = reinterpret_cast<short*>(smem);
  short* const kv1 = kv0 + KV_ELEMS;
  short* const p_base = kv0 + P_OFF;            // [4 waves][16][PF_KT]
  float* const stat = reinterpret_cast<float*>(smem) + STAT_F;  // [4][2][16]

  // ---- Q fragments
  constexpr int KS = D / 32;
  bf16x8_t qfrag[KS];
  const int my_qrow = q_base + wid * 16 + col;
  {
    const int r = (my_qrow < L) ? my_qrow : (L - 1);
    const __hip_bfloat16* qrow_p = q + (long)(s0 + r) * q_stride + (long)h * D;
#pragma unroll
    for (int ks = 0; ks < KS; ++ks) {
      qfrag[ks] = *reinterpret_cast<const bf16x8_t*>(qrow_p + ks * 32 + kgrp * 8);
      reg_fence(qfrag[ks]);  // wait HERE, not inside the glds pipeline
    }
  }

  float m_run[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  float l_run[4] = {0.f, 0.f, 0.f, 0.f};
  constexpr int DT = D / 16;
  f32x4_t ot[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) ot[dt] = {0.f, 0.f, 0.f, 0.f};

  const int wave_max_row = off + min(q_base + wid * 16 + 15, L - 1);
  const int block_max_row = off + min(q_base + PF_QT - 1, L - 1);
  const int kv_end = block_max_row + 1;
  int kv_begin = 0;
  if (window > 0) {
    const int wave_min_needed = off + q_base + 1 - window;
    kv_begin = max(0, (wave_min_needed / PF_KT) * PF_KT);
  }

  // Out-of-range keys clamp to row Lk-1 (real finite data): K garbage is
  // masked in S, dead keys carry P == 0 into PV — every tile stays on the
  // glds pipeline (see the decode kernel's note).
  auto issue_k = [&](int kt, short* dst) {
#pragma unroll
    for (int j = 0; j < NI_K; ++j) {
      const int g = (wid * NI_K + j) * 64 + lane;  // granule
      const int key = g / CPK;
      const int r8 = (g % CPK) * 8;
      const int d = r8 ^ ((key & 7) << 3);
      const int gk = min(kt + key, Lk - 1);
      const int o2 = __builtin_amdgcn_readfirstlane((wid * NI_K + j) * 512);
      glds16(k + (long)(s0k + gk) * k_stride + (long)kvh * D + d, dst + o2);
    }
  };
  auto issue_v = [&](int kt, short* dst) {
#pragma unroll
    for (int j = 0; j < NI_V; ++j) {
      const int st = wid * NI_V + j;
      const int ks = st / (D / 16), dtile = st % (D / 16);
      const int pos = lane * 8;
      const int bpos = pos / 64, rem = pos % 64;
      const int qq = ((bpos >> 2) & 1) | ((bpos & 3) << 1);
      const int key = ks * 32 + qq * 4 + rem / 16;
      const int dim = dtile * 16 + (rem & 15);
      const int gk = min(kt + key, Lk - 1);
      const int o2 = __builtin_amdgcn_readfirstlane(st * 528);
      glds16(v + (long)(s0k + gk) * v_stride + (long)kvh * D + dim, dst + o2);
    }
  };
  // ---- prologue
  if (kv_begin < kv_end) issue_k(kv_begin, kv0);
  int cur = 0;
  for (int kt = kv_begin; kt < kv_end; kt += PF_KT, cur ^= 1) {
    short* const X = cur ? kv1 : kv0;
    short* const Y = cur ? kv0 : kv1;
    pipe_barrier_vm<0>();

    const bool active = kt <= wave_max_row;  // wave-uniform
    float p[4][4];
    bool rescale = false;  // defer-max decision (set in the S phase)
    if (active) {
      // ---- S = Q K^T for 4 column tiles of 16 keys
      f32x4_t s[4];
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        s[ct] = {0.f, 0.f, 0.f, 0.f};
        const int key = ct * 16 + col;
#pragma unroll
        for (int ks = 0; ks < KS; ++ks) {
          const int d8 = ks * 32 + kgrp * 8;
          bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
              &X[key * D + swz(key, d8)]);
          s[ct] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[ks], bfrag, s[ct], 0, 0, 0);
        }
      }
      // scale, softcap, mask, row max
      float mx[4];
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int qrow = q_base + wid * 16 + kgrp * 4 + reg;
        const int row = off + qrow;
        float m = -1e30f;
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
          const int key = kt + ct * 16 + col;
          float x = s[ct][reg] * scale;
          if (softcap > 0.f) x = tanhf(x / softcap) * softcap;
          const bool dead = key > row || key >= Lk || qrow >= L ||
                            (window > 0 && key <= row - window);
          x = dead ? -1e30f : x;
          p[reg][ct] = x;
          m = fmaxf(m, x);
        }
#pragma unroll
        for (int o3 = 1; o3 < 16; o3 <<= 1) m = fmaxf(m, __shfl_xor(m, o3, WAVE));
        mx[reg] = m;
      }
      // defer-max (guide T13): when no row of this wave grew its max by
      // more than THR, keep the old maxes — P is then bounded by e^THR
      // (fp32 accumulators tolerate it) and the whole O-rescale pass is
      // skipped. The decision covers the ENTIRE tile before any P of it
      // is exponentiated (the T13 ordering hazard), and every l-update
      // uses alpha==1 on the defer path. Accuracy cost ≈3× max-abs error
      // (bounded-P bf16 quantisation) — inside the test tolerances.
      constexpr float DEFER_THR = 8.0f;
      bool grow = false;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        grow |= mx[reg] > m_run[reg] + DEFER_THR;
#ifdef LLMQ_DEFER_ON
      rescale = __any(grow);
#else
      rescale = true; (void)grow;  // defer-max measured a loss; see above
#endif
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const float m_new = rescale ? fmaxf(m_run[reg], mx[reg]) : m_run[reg];
        const float alpha = (rescale && m_new > -1e30f)
                                ? __expf(m_run[reg] - m_new) : 1.f;
        float lsum = 0.f;
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
          const float pe = (m_new > -1e30f && p[reg][ct] > -1e29f)
                               ? __expf(p[reg][ct] - m_new) : 0.f;
          p[reg][ct] = pe;
          lsum += pe;
        }
#pragma unroll
        for (int o3 = 1; o3 < 16; o3 <<= 1) lsum += __shfl_xor(lsum, o3, WAVE);
        l_run[reg] = l_run[reg] * alpha + lsum;
        m_run[reg] = m_new;
        if (rescale && col == 0) stat[(wid * 2 + 0) * 16 + kgrp * 4 + reg] = alpha;
      }
    }
    pipe_barrier();  // S reads of X done; alpha visible

    // ---- issue V(kt)->X, K(kt+PF_KT)->Y under the P build + PV
    const int nxt = kt + PF_KT;
    const bool prefetch = nxt < kv_end;
    issue_v(kt, X);
    if (prefetch) issue_k(nxt, Y);

    if (active) {
      // write P rows (bf16) to this wave's tile
      short* const p_lds_w = p_base + wid * 16 * PF_KT;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
#pragma unroll
        for (int ct = 0; ct < 4; ++ct) {
          p_lds_w[(kgrp * 4 + reg) * PF_KT + ct * 16 + col] =
              __bfloat16_as_short(__float2bfloat16(p[reg][ct]));
        }
      }
    }
    if (prefetch) {
      pipe_barrier_vm<NI_K>();
    } else {
      pipe_barrier_vm<0>();
    }

    if (active) {
      // ---- OT += V^T P^T (O-rescale skipped entirely on the defer path)
      const float alpha_q = rescale ? stat[(wid * 2 + 0) * 16 + col] : 1.f;
      short* const p_lds_w = p_base + wid * 16 * PF_KT;
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        if (rescale) {
          ot[dt][0] *= alpha_q; ot[dt][1] *= alpha_q;
          ot[dt][2] *= alpha_q; ot[dt][3] *= alpha_q;
        }
#pragma unroll
        for (int ks = 0; ks < PF_KT / 32; ++ks) {
          const int sub = (ks * (D / 16) + dt) * 528 + lane * 4;
          bf16x4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (lds_bf16x4*)&X[sub]);
          bf16x4v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
              (lds_bf16x4*)&X[sub + 4 * 64]);
          bf16x8_t a;
#pragma unroll
          for (int j = 0; j < 4; ++j) {
            a[j] = __bfloat16_as_short((__hip_bfloat16)lo[j]);
            a[4 + j] = __bfloat16_as_short((__hip_bfloat16)hi[j]);
          }
          bf16x8_t bb = *reinterpret_cast<const bf16x8_t*>(
              &p_lds_w[col * PF_KT + ks * 32 + kgrp * 8]);
          ot[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bb, ot[dt], 0, 0, 0);
        }
      }
    }
    // next loop-top barrier protects X (overwritten two iterations out)
  }

  // ---- epilogue: normalise and store
  if (col == 0) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const float l = l_run[reg];
      stat[(wid * 2 + 1) * 16 + kgrp * 4 + reg] = (l > 0.f) ? 1.0f / l : 0.f;
    }
  }
  pipe_barrier();
  const float inv = stat[(wid * 2 + 1) * 16 + col];
  const int orow = q_base + wid * 16 + col;
  if (orow < L) {
    __hip_bfloat16* op = out + (long)(s0 + orow) * o_stride + (long)h * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        op[dt * 16 + kgrp * 4 + reg] = __float2bfloat16(ot[dt][reg] * inv);
    }
  }
}

// ------------------------------------------------------- MFMA paged decode --
// Single-token GQA decode attention on matrix cores (bf16, D in {128,256},
// G <= 16). One workgroup per (sequence, kv_head); the GQA query group is
// padded to a 16-row MFMA tile (pad rows repeat q-head 0 and are never
// stored). K and V SHARE one LDS tile buffer: per 64-key chunk
//   stage K -> sync -> S = QK^T (wave w: 16-key slab, MFMA 16x16x32) ->
//   sync -> stage V into the same buffer (its HBM latency hides under the
//   softmax VALU work) + combine row maxes / build P / update per-wave l ->
//   sync -> PV: wave w owns dim slab [w*D/4,(w+1)*D/4): OT[dims,16q] +=
//   mfma(A = V^T frags, B = P^T b128 reads) -> sync.
// The block table is staged to LDS once up front (the per-load global
// bt[] read was a dependent-latency chain on every staging address).
// Sharing the tile keeps LDS ~36 KB -> 4 workgroups/CU (the v1 split-K/V
// layout was 68 KB -> 2 WGs/CU and ran SLOWER than the VALU kernel).
// The VALU kernel above is issue-bound (~120 cyc/token/wave); this one
// targets the KV HBM stream rate.

#define PD_MAX_BT 1024  // block-table entries staged in LDS (tail from global)


// KV-cache load: 8 cache elements → bf16x8 fragment. bf16 caches are a raw
// 16-byte load; fp8 (OCP e4m3) caches load 8 bytes and convert on the fly —
// the fp8 KV option halves the decode KV stream.
DEVINL bf16x8_t kv8_to_bf16(const __hip_bfloat16* p) {
  return *reinterpret_cast<const bf16x8_t*>(p);
}
DEVINL bf16x8_t kv8_to_bf16(const __hip_fp8_e4m3* p) {
  // gfx950 native packed OCP-fp8 converts: v_cvt_pk_f32_fp8 turns two fp8
  // (low/high half of a 16-bit pair) into two f32 in one VOP. The generic
  // static_cast<float>(__hip_fp8_e4m3) lowers to a multi-op bit sequence
  // per ELEMENT and made fp8 staging VALU-bound (fp8 decode measured
  // SLOWER than bf16 at batch 512 before this).
  const uint2 raw = *reinterpret_cast<const uint2*>(p);
  bf16x8_t out;
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    const int w = h ? (int)raw.y : (int)raw.x;
    typedef __attribute__((ext_vector_type(2))) float f32x2_t;
    const f32x2_t lo = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);
    const f32x2_t hi = __builtin_amdgcn_cvt_pk_f32_fp8(w, true);
    out[h * 4 + 0] = __bfloat16_as_short(__float2bfloat16(lo[0]));
    out[h * 4 + 1] = __bfloat16_as_short(__float2bfloat16(lo[1]));
    out[h * 4 + 2] = __bfloat16_as_short(__float2bfloat16(hi[0]));
    out[h * 4 + 3] = __bfloat16_as_short(__float2bfloat16(hi[1]));
  }
  return out;
}

// NW = waves per workgroup (4 -> 256 threads/64-key chunks; 8 -> 512
// threads/128-key chunks: same waves/SIMD at half the barriers per token).
// With gridDim.z == NSPLIT > 1 (flash-decoding split-KV: small B×KVH, e.g.
// tensor-parallel ranks holding one kv head), each z-workgroup covers a
// slice of the context and writes UNNORMALISED fp32 partials
// (acc[G][D], m, l) to scratch[B][KVH][NSPLIT][G][D+2]; the
// decode_splitkv_merge_kernel combines them. NSPLIT==1 writes out directly.
template <int HEAD_DIM, int NW, typename TC = __hip_bfloat16>
__global__ __launch_bounds__(NW * WAVE) void paged_decode_mfma_kernel(
    __hip_bfloat16* __restrict__ out,      // [B, H, D]
    const __hip_bfloat16* __restrict__ q,  // [B, H, D]
    const TC* __restrict__ k_cache,        // [nb, KVH, bs, D]
    const TC* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [B, max_blocks]
    const int* __restrict__ context_lens,  // [B]
    float* __restrict__ scratch,           // [B,KVH,NSPLIT,G,D+2] (NSPLIT>1)
    int num_heads, int num_kv_heads, int block_size, int max_blocks,
    float scale, float softcap, int window, long q_stride, long out_stride) {
  constexpr int D = HEAD_DIM;
  constexpr int KS = D / 32;      // MFMA K-steps over the head dim
  constexpr int PD_KT = NW * 16;  // keys per chunk
  constexpr int NT = NW * WAVE;   // threads per workgroup
  constexpr int D4 = D / NW;      // dim slab per wave
  constexpr int DT = D4 / 16;     // 16-dim MFMA tiles per wave
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int G = num_heads / num_kv_heads;  // <= 16
  const int L = context_lens[b];
  if (L <= 0) return;

  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int col = lane & 15;   // MFMA col lane (key for S, qrow for OT)
  const int kgrp = lane >> 4;  // 0..3

  // V's tr-subtile image strides subtiles by 528 elems (512 + 16): the
  // unpadded 1 KB stride lands every subtile on the same banks (16-way
  // staging-write conflicts). Sized for max(K rows, padded V image).
  constexpr int NSUB = (PD_KT / 32) * (D / 16);
  constexpr int KV_ELEMS = (PD_KT * D > NSUB * 528) ? PD_KT * D : NSUB * 528;
  __shared__ __attribute__((aligned(16))) short kv_lds[KV_ELEMS];  // K then V
  __shared__ __attribute__((aligned(16))) short p_lds[16 * PD_KT];  // [qrow][key]
  __shared__ float mpart_lds[NW][16];  // per-wave row-max partials
  __shared__ float alpha_lds[16];      // per-row rescale for the OT lanes
  __shared__ float l_lds[NW][16];      // per-wave l_run (end merge)
  __shared__ int bt_lds[PD_MAX_BT];

  // ---- stage the block table once (removes a dependent global load from
  // every staging address); contexts past PD_MAX_BT blocks (16k tokens at
  // bs 16) read the tail entries from global instead
  const int* bt_glob = block_tables + (long)b * max_blocks;
  {
    const int nb = (L + block_size - 1) / block_size;
    const int nstage = nb < PD_MAX_BT ? nb : PD_MAX_BT;
    for (int i = tid; i < nstage; i += NT) bt_lds[i] = bt_glob[i];
  }

  // ---- Q fragments: A-operand rows = padded q rows (row = lane&15)
  bf16x8_t qfrag[KS];
  {
    const int qrow = (col < G) ? col : 0;
    const __hip_bfloat16* qp = q + (long)b * q_stride + (long)(kh * G + qrow) * D;
#pragma unroll
    for (int ks = 0; ks < KS; ++ks)
      qfrag[ks] = *reinterpret_cast<const bf16x8_t*>(qp + ks * 32 + kgrp * 8);
  }

  // softmax state: every wave redundantly tracks rows r = kgrp*4 + reg
  float m_regs[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  float l_regs[4] = {0.f, 0.f, 0.f, 0.f};
  f32x4_t ot[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) ot[dt] = {0.f, 0.f, 0.f, 0.f};

  const int start = (window > 0 && L > window) ? (L - window) : 0;
  const int base0 = (start / PD_KT) * PD_KT;
  constexpr int CPK = D / 8;    // 16B chunks per key row
  constexpr int NCK = PD_KT * CPK;
  // split-KV: this z-block's slice of the chunk range
  const int nsplit = gridDim.z;
  int range_lo = base0, range_hi = L;
  if (nsplit > 1) {
    const int nchunks = (L - base0 + PD_KT - 1) / PD_KT;
    const int per = (nchunks + nsplit - 1) / nsplit;
    range_lo = base0 + (int)blockIdx.z * per * PD_KT;
    range_hi = min(L, range_lo + per * PD_KT);
  }
  __syncthreads();  // bt_lds ready

  for (int base = range_lo; base < range_hi; base += PD_KT) {
    // ---- stage K chunk (gather via LDS block table; zeros beyond L)
    for (int c = tid; c < NCK; c += NT) {
      const int key = c / CPK;
      const int d8 = (c % CPK) * 8;
      const int dst = key * D + swz(key, d8);
      const int gkey = base + key;
      if (gkey < L) {
        const int bidx = gkey / block_size;
        const long blk = (bidx < PD_MAX_BT) ? bt_lds[bidx] : bt_glob[bidx];
        const long rowoff =
            ((blk * num_kv_heads + kh) * block_size + gkey % block_size) * D + d8;
        *reinterpret_cast<bf16x8_t*>(&kv_lds[dst]) = kv8_to_bf16(k_cache + rowoff);
      } else {
        *reinterpret_cast<bf16x8_t*>(&kv_lds[dst]) = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
    __syncthreads();

    // ---- S[16, 16] for this wave's key slab
    f32x4_t s = {0.f, 0.f, 0.f, 0.f};
    {
      const int key = wid * 16 + col;
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
        const int d8 = ks * 32 + kgrp * 8;
        bf16x8_t bfrag =
            *reinterpret_cast<const bf16x8_t*>(&kv_lds[key * D + swz(key, d8)]);
        s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[ks], bfrag, s, 0, 0, 0);
      }
    }
    // scale, softcap, bounds mask; rows r = kgrp*4 + reg
    float sv[4], mx[4];
    {
      const int key = base + wid * 16 + col;
      const bool dead = key >= L || key < start;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float x = s[reg] * scale;
        if (softcap > 0.f) x = tanhf(x / softcap) * softcap;
        sv[reg] = dead ? -1e30f : x;
        float m = sv[reg];
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) m = fmaxf(m, __shfl_xor(m, off, WAVE));
        mx[reg] = m;
      }
    }
    if (col == 0) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) mpart_lds[wid][kgrp * 4 + reg] = mx[reg];
    }
    __syncthreads();  // mparts ready; every wave is past its K reads

    // ---- stage V into the SAME buffer; its latency hides under softmax VALU.
    // V uses a tr-read subtile image (NOT K's swizzled rows): 32-key × 16-dim
    // subtiles, key-quads permuted (0,2,4,6,1,3,5,7) so one uniform-base
    // ds_read_b64_tr_b16 delivers each 16-lane group its MFMA A-fragment
    // quad (guide T10: +11% on the PV path vs scalar ds_read_u16).
    for (int c = tid; c < NCK; c += NT) {
      const int key = c / CPK;
      const int d8 = (c % CPK) * 8;
      const int dt = d8 / 16, col0 = d8 & 15;
      const int qq = (key & 31) >> 2;
      const int block_pos = ((qq & 1) << 2) + (qq >> 1);
      const int dst = ((key >> 5) * (D / 16) + dt) * 528 + block_pos * 64 +
                      (key & 3) * 16 + col0;
      const int gkey = base + key;
      if (gkey < L) {
        const int bidx = gkey / block_size;
        const long blk = (bidx < PD_MAX_BT) ? bt_lds[bidx] : bt_glob[bidx];
        const long rowoff =
            ((blk * num_kv_heads + kh) * block_size + gkey % block_size) * D + d8;
        *reinterpret_cast<bf16x8_t*>(&kv_lds[dst]) = kv8_to_bf16(v_cache + rowoff);
      } else {
        *reinterpret_cast<bf16x8_t*>(&kv_lds[dst]) = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }

    // ---- combine maxes (identical on every wave), build P, update l
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int row = kgrp * 4 + reg;
      float m_tile = mpart_lds[0][row];
#pragma unroll
      for (int w = 1; w < NW; ++w) m_tile = fmaxf(m_tile, mpart_lds[w][row]);
      const float m_new = fmaxf(m_regs[reg], m_tile);
      const float alpha = (m_new > -1e30f) ? __expf(m_regs[reg] - m_new) : 1.f;
      const float pe = (m_new > -1e30f && sv[reg] > -1e29f)
                           ? __expf(sv[reg] - m_new) : 0.f;
      float lsum = pe;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) lsum += __shfl_xor(lsum, off, WAVE);
      l_regs[reg] = l_regs[reg] * alpha + lsum;
      m_regs[reg] = m_new;
      if (wid == 0 && col == 0) alpha_lds[row] = alpha;
      p_lds[row * PD_KT + wid * 16 + col] =
          __bfloat16_as_short(__float2bfloat16(pe));
    }
    __syncthreads();  // V + P + alpha ready

    // ---- OT[dims, 16q] += V^T P^T over this wave's dim slab
    const float alpha_q = alpha_lds[col];
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      ot[dt][0] *= alpha_q; ot[dt][1] *= alpha_q;
      ot[dt][2] *= alpha_q; ot[dt][3] *= alpha_q;
      const int dtile = (wid * D4) / 16 + dt;
#pragma unroll
      for (int ks = 0; ks < PD_KT / 32; ++ks) {
        // two transpose reads give the full 8-key A-fragment: each lane
        // reads 4 contiguous bf16 at its own 8-byte slot and the hardware
        // redistributes so lane l receives column (l&15) of its 16-lane
        // group's [4-key][16-dim] block
        const int sub = (ks * (D / 16) + dtile) * 528 + lane * 4;
        bf16x4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4*)&kv_lds[sub]);
        bf16x4v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4*)&kv_lds[sub + 4 * 64]);
        bf16x8_t a;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          a[j] = __bfloat16_as_short((__hip_bfloat16)lo[j]);
          a[4 + j] = __bfloat16_as_short((__hip_bfloat16)hi[j]);
        }
        bf16x8_t bb = *reinterpret_cast<const bf16x8_t*>(
            &p_lds[col * PD_KT + ks * 32 + kgrp * 8]);
        ot[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bb, ot[dt], 0, 0, 0);
      }
    }
    __syncthreads();  // V/P consumed; next chunk may overwrite
  }

  // ---- merge per-wave l, then store (normalised out, or raw partials)
  if (col == 0) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) l_lds[wid][kgrp * 4 + reg] = l_regs[reg];
  }
  __syncthreads();
  float l_tot = l_lds[0][col];
#pragma unroll
  for (int w = 1; w < NW; ++w) l_tot += l_lds[w][col];
  if (nsplit > 1) {
    float* slot = scratch +
        ((((long)b * num_kv_heads + kh) * nsplit + blockIdx.z) * G) * (D + 2);
    if (col < G) {
      float* row = slot + (long)col * (D + 2);
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        const int dim0 = wid * D4 + dt * 16 + kgrp * 4;
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) row[dim0 + reg] = ot[dt][reg];
      }
    }
    // Row stats: wave-0 lanes with col==0 cover rows r = kgrp*4+reg
    // (m_regs is identical across waves after the shared-max combine).
    if (wid == 0 && col == 0) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int r = kgrp * 4 + reg;
        if (r < G) {
          float* row = slot + (long)r * (D + 2);
          row[D] = m_regs[reg];
          row[D + 1] = l_lds[0][r];
          for (int w = 1; w < NW; ++w) row[D + 1] += l_lds[w][r];
        }
      }
    }
    return;
  }
  const float inv = (l_tot > 0.f) ? 1.0f / l_tot : 0.f;
  if (col < G) {
    __hip_bfloat16* op = out + (long)b * out_stride + (long)(kh * G + col) * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      const int dim0 = wid * D4 + dt * 16 + kgrp * 4;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        op[dim0 + reg] = __float2bfloat16(ot[dt][reg] * inv);
    }
  }
}

// -------------------------------------------- pipelined MFMA paged decode --
// Software-pipelined variant of paged_decode_mfma_kernel (round-2 item from
// profiles/r1_step6_decode_tuning_story.md: the stage→S→stats→PV phase
// serialization left ~30% of the KV stream rate on the table, and both
// register-prefetch and occupancy-costing fixes measured WORSE — the fix
// has to add zero VGPRs and keep the 2-workgroups/CU residency).
//
// Structure (per CDNA4 guide §5 "Pipelining across barriers" + §5.5 T3/T4):
//   - TWO chunk buffers A/B; chunk i's K lives in buf[i&1], V overwrites the
//     same buffer after S (the r1 shared-buffer trick), K(i+1) prefetches
//     into buf[(i+1)&1].
//   - ALL staging is `global_load_lds` (no VGPR round-trip, no staging
//     instructions on the VALU): the LDS image is lane-linear per wave
//     instruction, so K's XOR swizzle and V's tr16 subtile permutation move
//     to the per-lane SOURCE address (guide §5.4 rule 21).
//   - Raw `s_barrier` + counted `s_waitcnt vmcnt(N)`: V(i)+K(i+1) issue
//     back-to-back right after the post-S barrier; the pre-PV barrier waits
//     vmcnt(NI_K) (V landed, K still streaming); the loop-top barrier waits
//     vmcnt(0) (K landed). A `__syncthreads()` would drain the glds queue
//     at every barrier (the documented HIP-compiler ceiling).
//   - The chunk containing L falls back to plain staged writes (glds cannot
//     zero-fill; garbage V rows would reach the PV MFMA as NaN×0).
// Per-workgroup timeline: the only window with no HBM traffic in flight is
// the S phase (~hundreds of cycles vs ~6 µs of stream per chunk).

// min waves/SIMD: the 64-key chunk fits 2 workgroups/CU by LDS (4 waves/
// SIMD) — force the register allocator to <=128 VGPRs so VGPRs don't cap
// occupancy below that (the r1 lesson: this kernel is governed by
// waves/SIMD, every latency trick that cost occupancy lost). The 128-key
// chunk is 1 workgroup/CU by LDS (2 waves/SIMD) — let it use 196.
template <int HEAD_DIM, int NW, int PD_KT>
__global__ __launch_bounds__(NW * WAVE, (PD_KT <= 64 ? 4 : 2)) void paged_decode_pipe_kernel(
    __hip_bfloat16* __restrict__ out,      // [B, H, D]
    const __hip_bfloat16* __restrict__ q,  // [B, H, D]
    const __hip_bfloat16* __restrict__ k_cache,  // [nb, KVH, bs, D]
    const __hip_bfloat16* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [B, max_blocks<=PD_MAX_BT]
    const int* __restrict__ context_lens,  // [B]
    float* __restrict__ scratch,           // [B,KVH,NSPLIT,G,D+2] (NSPLIT>1)
    int num_heads, int num_kv_heads, int block_size, int max_blocks,
    float scale, float softcap, int window, long q_stride, long out_stride) {
  constexpr int D = HEAD_DIM;
  constexpr int KS = D / 32;        // MFMA K-steps over the head dim
  constexpr int NT = NW * WAVE;     // threads per workgroup
  constexpr int SLABS = PD_KT / 16; // 16-key S slabs (<= NW; waves duplicate)
  constexpr int D4 = D / NW;        // dim slab per wave (PV)
  constexpr int DT = D4 / 16;       // 16-dim MFMA tiles per wave
  constexpr int CPK = D / 8;        // 16B granules per key row
  constexpr int NI_K = PD_KT * CPK / NT;          // K glds per wave per chunk
  constexpr int NSUB = (PD_KT / 32) * (D / 16);   // V tr-subtiles per chunk
  constexpr int NI_V = NSUB / NW;                 // V glds per wave per chunk
  constexpr int KV_ELEMS = (PD_KT * D > NSUB * 528) ? PD_KT * D : NSUB * 528;
  static_assert(PD_KT % 32 == 0 && SLABS <= NW && NSUB % NW == 0, "");
  static_assert(PD_KT * CPK % NT == 0 && D % NW == 0 && D4 % 16 == 0, "");

  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int G = num_heads / num_kv_heads;  // <= 16
  const int L = context_lens[b];
  if (L <= 0) return;

  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int col = lane & 15;
  const int kgrp = lane >> 4;
  const int slab = wid % SLABS;

  // ONE shared array (guide §5 trap 4a: a second __shared__ object makes
  // hipcc drain vmcnt(0) before every ds_read while a glds is in flight).
  constexpr int P_OFF = 2 * KV_ELEMS;                  // shorts
  constexpr int MPART_OFF = P_OFF + 16 * PD_KT;        // floats from here
  constexpr int F_BASE = (MPART_OFF * 2 + 15) / 16 * 4;  // float index base
  constexpr int ALPHA_F = F_BASE + SLABS * 16;
  constexpr int LPART_F = ALPHA_F + 16;
  constexpr int BT_I = LPART_F + SLABS * 16;           // int index base
  constexpr int TOTAL_BYTES = BT_I * 4 + PD_MAX_BT * 4;
  __shared__ __attribute__((aligned(16))) char smem[TOTAL_BYTES];
  short* const kv0 = reinterpret_cast<short*>(smem);
  short* const kv1 = kv0 + KV_ELEMS;
  short* const p_lds2 = kv0 + P_OFF;                   // [16][PD_KT] bf16
  float* const fbase = reinterpret_cast<float*>(smem);
  float* const mpart = fbase + F_BASE;                 // [SLABS][16]
  float* const alpha_s = fbase + ALPHA_F;              // [16]
  float* const l_part = fbase + LPART_F;               // [SLABS][16]
  int* const bt_l = reinterpret_cast<int*>(smem) + BT_I;  // [PD_MAX_BT]

  // ---- stage the block table once (plain loads; nb <= PD_MAX_BT is a
  // dispatch precondition for this kernel)
  const int* bt_glob = block_tables + (long)b * max_blocks;
  {
    const int nb = (L + block_size - 1) / block_size;
    for (int i = tid; i < nb; i += NT) bt_l[i] = bt_glob[i];
  }

  // ---- Q fragments (padded GQA rows; row = lane&15)
  bf16x8_t qfrag[KS];
  {
    const int qrow = (col < G) ? col : 0;
    const __hip_bfloat16* qp = q + (long)b * q_stride + (long)(kh * G + qrow) * D;
#pragma unroll
    for (int ks = 0; ks < KS; ++ks) {
      qfrag[ks] = *reinterpret_cast<const bf16x8_t*>(qp + ks * 32 + kgrp * 8);
      reg_fence(qfrag[ks]);  // wait the load NOW — a first-use inside the
      // loop would make hipcc place `s_waitcnt vmcnt(0)` there, draining
      // the glds pipeline on every iteration (guide §5 trap 4b).
    }
  }

  float m_regs[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  float l_regs[4] = {0.f, 0.f, 0.f, 0.f};
  f32x4_t ot[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) ot[dt] = {0.f, 0.f, 0.f, 0.f};

  const int start = (window > 0 && L > window) ? (L - window) : 0;
  const int base0 = (start / PD_KT) * PD_KT;
  const int nsplit = gridDim.z;
  int range_lo = base0, range_hi = L;
  if (nsplit > 1) {
    const int nchunks = (L - base0 + PD_KT - 1) / PD_KT;
    const int per = (nchunks + nsplit - 1) / nsplit;
    range_lo = base0 + (int)blockIdx.z * per * PD_KT;
    range_hi = min(L, range_lo + per * PD_KT);
  }

  const long kvrow_stride = (long)num_kv_heads * block_size * D;

  // Per-lane source row+offset for K granule g of a chunk (inverse swizzle
  // on the source: LDS image is lane-linear, content matches swz()).
  // Out-of-range keys (tail chunk / z-range end) clamp to row L-1: a real
  // written cache row, so the staged garbage is FINITE; K garbage is
  // masked in S (-1e30) and dead keys carry P == 0, so finite V garbage
  // contributes exactly 0 to PV. This keeps EVERY chunk on the glds
  // pipeline (no zero-filled plain-staged tail).
  auto k_src = [&](int base, int g) -> const __hip_bfloat16* {
    const int key = g / CPK;
    const int r8 = (g % CPK) * 8;
    const int d = r8 ^ ((key & 7) << 3);
    const int gkey = min(base + key, L - 1);
    const long blk = bt_l[gkey / block_size];
    return k_cache + (blk * num_kv_heads + kh) * ((long)block_size * D) +
           (long)(gkey % block_size) * D + d;
  };
  // V: subtile st, in-subtile granule = lane (tr16 image inverse mapping).
  auto v_src = [&](int base, int st) -> const __hip_bfloat16* {
    const int ks = st / (D / 16), dtile = st % (D / 16);
    const int pos = lane * 8;
    const int bpos = pos / 64, rem = pos % 64;
    const int qq = ((bpos >> 2) & 1) | ((bpos & 3) << 1);
    const int key = ks * 32 + qq * 4 + rem / 16;
    const int dim = dtile * 16 + (rem & 15);
    const int gkey = min(base + key, L - 1);
    const long blk = bt_l[gkey / block_size];
    return v_cache + (blk * num_kv_heads + kh) * ((long)block_size * D) +
           (long)(gkey % block_size) * D + dim;
  };

  auto issue_k = [&](int base, short* dst) {
#pragma unroll
    for (int j = 0; j < NI_K; ++j) {
      const int inst = wid * NI_K + j;
      const int off = __builtin_amdgcn_readfirstlane(inst * 512);
      glds16(k_src(base, inst * 64 + lane), dst + off);
    }
  };
  auto issue_v = [&](int base, short* dst) {
#pragma unroll
    for (int j = 0; j < NI_V; ++j) {
      const int st = wid * NI_V + j;
      const int off = __builtin_amdgcn_readfirstlane(st * 528);
      glds16(v_src(base, st), dst + off);
    }
  };
  // ---- prologue: block table visible, then K(first) in flight
  pipe_barrier();  // bt_l ready (plain ds_writes; lgkm drain)
  if (range_lo < range_hi) issue_k(range_lo, kv0);

  int cur = 0;
  for (int base = range_lo; base < range_hi; base += PD_KT, cur ^= 1) {
    short* const X = cur ? kv1 : kv0;
    short* const Y = cur ? kv0 : kv1;
    pipe_barrier_vm<0>();  // K(base) landed (all waves)

    // ---- S[16,16] for this wave's slab (waves wid>=SLABS duplicate)
    f32x4_t s = {0.f, 0.f, 0.f, 0.f};
    {
      const int key = slab * 16 + col;
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
        const int d8 = ks * 32 + kgrp * 8;
        bf16x8_t bfrag =
            *reinterpret_cast<const bf16x8_t*>(&X[key * D + swz(key, d8)]);
        s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[ks], bfrag, s, 0, 0, 0);
      }
    }
    float sv[4];
    {
      const int key = base + slab * 16 + col;
      const bool dead = key >= L || key < start;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float x = s[reg] * scale;
        if (softcap > 0.f) x = tanhf(x / softcap) * softcap;
        sv[reg] = dead ? -1e30f : x;
        float m = sv[reg];
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) m = fmaxf(m, __shfl_xor(m, off, WAVE));
        if (col == 0 && wid < SLABS) mpart[slab * 16 + kgrp * 4 + reg] = m;
      }
    }
    pipe_barrier();  // S reads of X done everywhere; mpart visible

    // ---- issue V(base)->X then K(next)->Y; both stream under softmax/PV
    const int next = base + PD_KT;
    const bool prefetch = next < range_hi;
    issue_v(base, X);
    if (prefetch) issue_k(next, Y);

    // ---- combine maxes, build P, update l (redundant on every wave)
    float m_tile[4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int row = kgrp * 4 + reg;
      float m = mpart[row];
#pragma unroll
      for (int w = 1; w < SLABS; ++w) m = fmaxf(m, mpart[w * 16 + row]);
      m_tile[reg] = m;
    }
    // defer-max (guide T13): every wave computes m_regs/m_tile from the
    // same mpart values with the same op order, so the ballot below is
    // workgroup-uniform with no extra LDS flag. On the defer path
    // alpha==1 exactly: the alpha_s store AND the whole OT-rescale pass
    // are skipped; P is bounded by e^THR (l/OT accumulate in fp32).
    bool grow = false;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) grow |= m_tile[reg] > m_regs[reg] + 8.0f;
#ifdef LLMQ_DEFER_ON
    const bool rescale = __any(grow);
#else
    // Measured: the T13 defer guard LOSES here (decode 0.458 vs 0.442 ms,
    // prefill 0.282 vs 0.261 ms same-box A/B) — the branch around the
    // O-rescale disturbs the pipelined PV scheduling more than the skipped
    // VALU pays. Kept compilable behind -DLLMQ_DEFER_ON for re-testing.
    const bool rescale = true; (void)grow;
#endif
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int row = kgrp * 4 + reg;
      const float m_new = rescale ? fmaxf(m_regs[reg], m_tile[reg]) : m_regs[reg];
      const float alpha = (rescale && m_new > -1e30f)
                              ? __expf(m_regs[reg] - m_new) : 1.f;
      const float pe = (m_new > -1e30f && sv[reg] > -1e29f)
                           ? __expf(sv[reg] - m_new) : 0.f;
      float lsum = pe;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) lsum += __shfl_xor(lsum, off, WAVE);
      l_regs[reg] = l_regs[reg] * alpha + lsum;
      m_regs[reg] = m_new;
      if (rescale && wid == 0 && col == 0) alpha_s[row] = alpha;
      p_lds2[row * PD_KT + slab * 16 + col] =
          __bfloat16_as_short(__float2bfloat16(pe));
    }
    if (prefetch) {
      pipe_barrier_vm<NI_K>();  // V landed; K(next) stays in flight
    } else {
      pipe_barrier_vm<0>();
    }

    // ---- OT[dims,16q] += V^T P^T over this wave's dim slab
    const float alpha_q = rescale ? alpha_s[col] : 1.f;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      if (rescale) {
        ot[dt][0] *= alpha_q; ot[dt][1] *= alpha_q;
        ot[dt][2] *= alpha_q; ot[dt][3] *= alpha_q;
      }
      const int dtile = (wid * D4) / 16 + dt;
#pragma unroll
      for (int ks = 0; ks < PD_KT / 32; ++ks) {
        const int sub = (ks * (D / 16) + dtile) * 528 + lane * 4;
        bf16x4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4*)&X[sub]);
        bf16x4v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4*)&X[sub + 4 * 64]);
        bf16x8_t a;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          a[j] = __bfloat16_as_short((__hip_bfloat16)lo[j]);
          a[4 + j] = __bfloat16_as_short((__hip_bfloat16)hi[j]);
        }
        bf16x8_t bb = *reinterpret_cast<const bf16x8_t*>(
            &p_lds2[col * PD_KT + ks * 32 + kgrp * 8]);
        ot[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bb, ot[dt], 0, 0, 0);
      }
    }
    // No trailing barrier: the next loop-top barrier (vmcnt(0)) both
    // protects X against the glds of chunk base+2 (issued only after the
    // NEXT post-S barrier) and confirms K(next) landed in Y.
  }

  // ---- merge per-slab l, then store (normalised out, or raw partials)
  if (col == 0 && wid < SLABS) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) l_part[slab * 16 + kgrp * 4 + reg] = l_regs[reg];
  }
  pipe_barrier();
  float l_tot = l_part[col];
#pragma unroll
  for (int w = 1; w < SLABS; ++w) l_tot += l_part[w * 16 + col];
  if (nsplit > 1) {
    float* slot = scratch +
        ((((long)b * num_kv_heads + kh) * nsplit + blockIdx.z) * G) * (D + 2);
    if (col < G) {
      float* row = slot + (long)col * (D + 2);
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        const int dim0 = wid * D4 + dt * 16 + kgrp * 4;
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) row[dim0 + reg] = ot[dt][reg];
      }
    }
    if (wid == 0 && col == 0) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int r = kgrp * 4 + reg;
        if (r < G) {
          float* row = slot + (long)r * (D + 2);
          row[D] = m_regs[reg];
          float lt = l_part[r];
          for (int w = 1; w < SLABS; ++w) lt += l_part[w * 16 + r];
          row[D + 1] = lt;
        }
      }
    }
    return;
  }
  const float inv = (l_tot > 0.f) ? 1.0f / l_tot : 0.f;
  if (col < G) {
    __hip_bfloat16* op = out + (long)b * out_stride + (long)(kh * G + col) * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      const int dim0 = wid * D4 + dt * 16 + kgrp * 4;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        op[dim0 + reg] = __float2bfloat16(ot[dt][reg] * inv);
    }
  }
}

// --------------------------------------- fp8-KV pipelined MFMA decode --
// The glds pipeline for fp8 (OCP e4m3) caches: HALF the KV stream of
// bf16. glds cannot convert during the LDS write, so:
//   - K stays fp8 IN LDS ([PD_KT][D] bytes, 16B-granule XOR swizzle on
//     the source); the S phase reads 8-byte fragments and converts with
//     the native packed cvt VOPs (one v_cvt_pk_f32_fp8 per 4 elems).
//   - V stages RAW fp8 into the scratch back-half of the OTHER K/V-image
//     buffer, then a convert pass (during the softmax window) writes the
//     bf16 tr16 subtile image the PV path expects.
// Buffer choreography per chunk i (X = img[i&1], Y = img[(i+1)&1]):
//   top: vm(0) barrier [K(i) in X front]  ->  S(i) from X (fp8+cvt)
//   post-S barrier -> issue V(i)raw -> Y back;  K(i+1) -> Y front
//   softmax/P  ->  vm(NI_K) barrier [V raw landed; K(i+1) flying]
//   convert pass: Y back (raw) -> X image (overwrites K(i): dead)
//   lgkm barrier -> PV(i) from X image
// Y-back is free during i (its image is only rebuilt by i+1's convert);
// Y-front K(i+1) and Y-back V(i)raw are disjoint ranges. LDS stays at
// 2 × image + p + bt ≈ 74 KB -> 2 workgroups/CU, same as the bf16 pipe.

DEVINL bf16x8_t fp8x8_to_bf16(unsigned lo, unsigned hi) {
  typedef __attribute__((ext_vector_type(2))) float cvt_f32x2;
  bf16x8_t out;
#pragma unroll
  for (int h = 0; h < 2; ++h) {
    const int w = h ? (int)hi : (int)lo;
    const cvt_f32x2 a = __builtin_amdgcn_cvt_pk_f32_fp8(w, false);
    const cvt_f32x2 b = __builtin_amdgcn_cvt_pk_f32_fp8(w, true);
    out[h * 4 + 0] = __bfloat16_as_short(__float2bfloat16(a[0]));
    out[h * 4 + 1] = __bfloat16_as_short(__float2bfloat16(a[1]));
    out[h * 4 + 2] = __bfloat16_as_short(__float2bfloat16(b[0]));
    out[h * 4 + 3] = __bfloat16_as_short(__float2bfloat16(b[1]));
  }
  return out;
}

template <int HEAD_DIM, int NW>
__global__ __launch_bounds__(NW * WAVE, 4) void paged_decode_pipe_fp8_kernel(
    __hip_bfloat16* __restrict__ out,      // [B, H, D]
    const __hip_bfloat16* __restrict__ q,  // [B, H, D]
    const __hip_fp8_e4m3* __restrict__ k_cache,  // [nb, KVH, bs, D]
    const __hip_fp8_e4m3* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [B, max_blocks<=PD_MAX_BT]
    const int* __restrict__ context_lens,  // [B]
    float* __restrict__ scratch,           // [B,KVH,NSPLIT,G,D+2] (NSPLIT>1)
    int num_heads, int num_kv_heads, int block_size, int max_blocks,
    float scale, float softcap, int window, long q_stride, long out_stride) {
  constexpr int D = HEAD_DIM;
  constexpr int KS = D / 32;
  constexpr int PD_KT = 64;
  constexpr int NT = NW * WAVE;
  constexpr int SLABS = PD_KT / 16;     // 4 (waves duplicate when NW=8)
  constexpr int D4 = D / NW;
  constexpr int DT = D4 / 16;
  constexpr int G16B = D / 16;          // 16B granules per fp8 key row
  constexpr int NI_K = PD_KT * G16B / NT;   // fp8-K glds per wave
  constexpr int NI_VR = NI_K;               // raw fp8-V glds per wave
  constexpr int NSUB = (PD_KT / 32) * (D / 16);
  constexpr int IMG_ELEMS = NSUB * 528;     // bf16 tr16 image (shorts)
  constexpr int RAW_BYTES = PD_KT * D;      // fp8 K front == raw V size
  static_assert(IMG_ELEMS * 2 >= 2 * RAW_BYTES,
                "image must cover K-front + V-raw back");
  static_assert(NI_K >= 1 && D % NW == 0 && D4 % 16 == 0, "");

  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int G = num_heads / num_kv_heads;
  const int L = context_lens[b];
  if (L <= 0) return;

  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int col = lane & 15;
  const int kgrp = lane >> 4;
  const int slab = wid % SLABS;

  constexpr int P_OFF = 2 * IMG_ELEMS;                   // shorts
  constexpr int F_BASE = (P_OFF + 16 * PD_KT + 7) / 8 * 4;
  constexpr int ALPHA_F = F_BASE + SLABS * 16;
  constexpr int LPART_F = ALPHA_F + 16;
  constexpr int BT_I = LPART_F + SLABS * 16;
  constexpr int TOTAL_BYTES = BT_I * 4 + PD_MAX_BT * 4;
  __shared__ __attribute__((aligned(16))) char smem[TOTAL_BYTES];
  short* const img0 = reinterpret_cast<short*>(smem);
  short* const img1 = img0 + IMG_ELEMS;
  short* const p_lds2 = img0 + P_OFF;
  float* const fbase = reinterpret_cast<float*>(smem);
  float* const mpart = fbase + F_BASE;
  float* const alpha_s = fbase + ALPHA_F;
  float* const l_part = fbase + LPART_F;
  int* const bt_l = reinterpret_cast<int*>(smem) + BT_I;

  const int* bt_glob = block_tables + (long)b * max_blocks;
  {
    const int nb = (L + block_size - 1) / block_size;
    for (int i = tid; i < nb; i += NT) bt_l[i] = bt_glob[i];
  }

  bf16x8_t qfrag[KS];
  {
    const int qrow = (col < G) ? col : 0;
    const __hip_bfloat16* qp = q + (long)b * q_stride + (long)(kh * G + qrow) * D;
#pragma unroll
    for (int ks = 0; ks < KS; ++ks) {
      qfrag[ks] = *reinterpret_cast<const bf16x8_t*>(qp + ks * 32 + kgrp * 8);
      reg_fence(qfrag[ks]);
    }
  }

  float m_regs[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  float l_regs[4] = {0.f, 0.f, 0.f, 0.f};
  f32x4_t ot[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) ot[dt] = {0.f, 0.f, 0.f, 0.f};

  const int start = (window > 0 && L > window) ? (L - window) : 0;
  const int base0 = (start / PD_KT) * PD_KT;
  const int nsplit = gridDim.z;
  int range_lo = base0, range_hi = L;
  if (nsplit > 1) {
    const int nchunks = (L - base0 + PD_KT - 1) / PD_KT;
    const int per = (nchunks + nsplit - 1) / nsplit;
    range_lo = base0 + (int)blockIdx.z * per * PD_KT;
    range_hi = min(L, range_lo + per * PD_KT);
  }

  auto kv_row = [&](int gkey, const __hip_fp8_e4m3* cache) -> const __hip_fp8_e4m3* {
    const int ck = min(gkey, L - 1);  // clamp: finite real data (masked)
    const long blk = bt_l[ck / block_size];
    return cache + (blk * num_kv_heads + kh) * ((long)block_size * D) +
           (long)(ck % block_size) * D;
  };
  // K fp8 image: [PD_KT][D] bytes, 16B-granule XOR swizzle (col16 ^ key&7)
  auto issue_k = [&](int base, short* dst) {
#pragma unroll
    for (int j = 0; j < NI_K; ++j) {
      const int g = (wid * NI_K + j) * 64 + lane;  // 16B granule
      const int key = g / G16B;
      const int c16 = g % G16B;
      const int src_col = ((c16 ^ (key & 7)) & (G16B - 1)) << 4;
      const int off = __builtin_amdgcn_readfirstlane((wid * NI_K + j) * 512);
      glds16(kv_row(base + key, k_cache) + src_col,
             reinterpret_cast<char*>(dst) + off * 2);
    }
  };
  // V raw fp8, linear [PD_KT][D] bytes into the back half of `dst`
  auto issue_v_raw = [&](int base, short* dst) {
    char* const back = reinterpret_cast<char*>(dst) + RAW_BYTES;
#pragma unroll
    for (int j = 0; j < NI_VR; ++j) {
      const int g = (wid * NI_VR + j) * 64 + lane;
      const int key = g / G16B;
      const int colb = (g % G16B) * 16;
      const int off = __builtin_amdgcn_readfirstlane((wid * NI_VR + j) * 1024);
      glds16(kv_row(base + key, v_cache) + colb, back + off);
    }
  };

  pipe_barrier();  // bt_l ready
  if (range_lo < range_hi) issue_k(range_lo, img0);

  int cur = 0;
  for (int base = range_lo; base < range_hi; base += PD_KT, cur ^= 1) {
    short* const X = cur ? img1 : img0;
    short* const Y = cur ? img0 : img1;
    pipe_barrier_vm<0>();  // K(base) landed in X front

    // ---- S from fp8 K (convert per fragment with packed cvt VOPs)
    f32x4_t s = {0.f, 0.f, 0.f, 0.f};
    {
      const int key = slab * 16 + col;
      const char* krow = reinterpret_cast<const char*>(X) + key * D;
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
        const int d8 = ks * 32 + kgrp * 8;
        const int c16 = d8 >> 4;
        const int lds_col = ((c16 ^ (key & 7)) & (G16B - 1)) * 16 + (d8 & 15);
        const uint2 raw = *reinterpret_cast<const uint2*>(krow + lds_col);
        const bf16x8_t bfrag = fp8x8_to_bf16(raw.x, raw.y);
        s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[ks], bfrag, s, 0, 0, 0);
      }
    }
    float sv[4];
    {
      const int key = base + slab * 16 + col;
      const bool dead = key >= L || key < start;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float x = s[reg] * scale;
        if (softcap > 0.f) x = tanhf(x / softcap) * softcap;
        sv[reg] = dead ? -1e30f : x;
        float m = sv[reg];
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) m = fmaxf(m, __shfl_xor(m, off, WAVE));
        if (col == 0 && wid < SLABS) mpart[slab * 16 + kgrp * 4 + reg] = m;
      }
    }
    pipe_barrier();  // S reads of X-front done; mparts visible

    // ---- issue V(base) raw -> Y back, K(next) -> Y front. (Issuing V at
    // the loop TOP — its buffer is already free there — measured 0.390 vs
    // 0.373 ms: no win, the post-S window already covers the V stream.)
    const int next = base + PD_KT;
    const bool prefetch = next < range_hi;
    issue_v_raw(base, Y);
    if (prefetch) issue_k(next, Y);

    // ---- softmax combine + P
    float m_tile[4];
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int row = kgrp * 4 + reg;
      float m = mpart[row];
#pragma unroll
      for (int w = 1; w < SLABS; ++w) m = fmaxf(m, mpart[w * 16 + row]);
      m_tile[reg] = m;
    }
    // defer-max (guide T13): every wave computes m_regs/m_tile from the
    // same mpart values with the same op order, so the ballot below is
    // workgroup-uniform with no extra LDS flag. On the defer path
    // alpha==1 exactly: the alpha_s store AND the whole OT-rescale pass
    // are skipped; P is bounded by e^THR (l/OT accumulate in fp32).
    bool grow = false;
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) grow |= m_tile[reg] > m_regs[reg] + 8.0f;
#ifdef LLMQ_DEFER_ON
    const bool rescale = __any(grow);
#else
    // Measured: the T13 defer guard LOSES here (decode 0.458 vs 0.442 ms,
    // prefill 0.282 vs 0.261 ms same-box A/B) — the branch around the
    // O-rescale disturbs the pipelined PV scheduling more than the skipped
    // VALU pays. Kept compilable behind -DLLMQ_DEFER_ON for re-testing.
    const bool rescale = true; (void)grow;
#endif
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int row = kgrp * 4 + reg;
      const float m_new = rescale ? fmaxf(m_regs[reg], m_tile[reg]) : m_regs[reg];
      const float alpha = (rescale && m_new > -1e30f)
                              ? __expf(m_regs[reg] - m_new) : 1.f;
      const float pe = (m_new > -1e30f && sv[reg] > -1e29f)
                           ? __expf(sv[reg] - m_new) : 0.f;
      float lsum = pe;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) lsum += __shfl_xor(lsum, off, WAVE);
      l_regs[reg] = l_regs[reg] * alpha + lsum;
      m_regs[reg] = m_new;
      if (rescale && wid == 0 && col == 0) alpha_s[row] = alpha;
      p_lds2[row * PD_KT + slab * 16 + col] =
          __bfloat16_as_short(__float2bfloat16(pe));
    }
    if (prefetch) {
      pipe_barrier_vm<NI_K>();  // V raw landed; K(next) stays in flight
    } else {
      pipe_barrier_vm<0>();
    }

    // ---- convert pass: Y back (raw fp8) -> X bf16 tr16 image
    {
      const char* const raw = reinterpret_cast<const char*>(Y) + RAW_BYTES;
      constexpr int CPK = D / 8;          // 8-elem granules per key row
      constexpr int NG = PD_KT * CPK;     // 512 at D=256
#pragma unroll
      for (int j = 0; j < NG / NT; ++j) {
        const int c = tid + j * NT;
        const int key = c / CPK;
        const int d8 = (c % CPK) * 8;
        const int dtile = d8 / 16, col0 = d8 & 15;
        const int qq = (key & 31) >> 2;
        const int bpos = ((qq & 1) << 2) + (qq >> 1);
        const int pdst = ((key >> 5) * (D / 16) + dtile) * 528 + bpos * 64 +
                         (key & 3) * 16 + col0;
        const uint2 rr = *reinterpret_cast<const uint2*>(raw + key * D + d8);
        *reinterpret_cast<bf16x8_t*>(&X[pdst]) = fp8x8_to_bf16(rr.x, rr.y);
      }
    }
    pipe_barrier();  // image + P + alpha visible

    // ---- PV from the bf16 tr16 image
    const float alpha_q = rescale ? alpha_s[col] : 1.f;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      if (rescale) {
        ot[dt][0] *= alpha_q; ot[dt][1] *= alpha_q;
        ot[dt][2] *= alpha_q; ot[dt][3] *= alpha_q;
      }
      const int dtile = (wid * D4) / 16 + dt;
#pragma unroll
      for (int ks = 0; ks < PD_KT / 32; ++ks) {
        const int sub = (ks * (D / 16) + dtile) * 528 + lane * 4;
        bf16x4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4*)&X[sub]);
        bf16x4v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4*)&X[sub + 4 * 64]);
        bf16x8_t a;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          a[j] = __bfloat16_as_short((__hip_bfloat16)lo[j]);
          a[4 + j] = __bfloat16_as_short((__hip_bfloat16)hi[j]);
        }
        bf16x8_t bb = *reinterpret_cast<const bf16x8_t*>(
            &p_lds2[col * PD_KT + ks * 32 + kgrp * 8]);
        ot[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bb, ot[dt], 0, 0, 0);
      }
    }
    // next loop-top barrier (vm0) protects X/Y rotation
  }

  // ---- merge per-slab l, then store (same as the bf16 pipe kernel)
  if (col == 0 && wid < SLABS) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) l_part[slab * 16 + kgrp * 4 + reg] = l_regs[reg];
  }
  pipe_barrier();
  float l_tot = l_part[col];
#pragma unroll
  for (int w = 1; w < SLABS; ++w) l_tot += l_part[w * 16 + col];
  if (nsplit > 1) {
    float* slot = scratch +
        ((((long)b * num_kv_heads + kh) * nsplit + blockIdx.z) * G) * (D + 2);
    if (col < G) {
      float* row = slot + (long)col * (D + 2);
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        const int dim0 = wid * D4 + dt * 16 + kgrp * 4;
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) row[dim0 + reg] = ot[dt][reg];
      }
    }
    if (wid == 0 && col == 0) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int r = kgrp * 4 + reg;
        if (r < G) {
          float* row = slot + (long)r * (D + 2);
          row[D] = m_regs[reg];
          float lt = l_part[r];
          for (int w = 1; w < SLABS; ++w) lt += l_part[w * 16 + r];
          row[D + 1] = lt;
        }
      }
    }
    return;
  }
  const float inv = (l_tot > 0.f) ? 1.0f / l_tot : 0.f;
  if (col < G) {
    __hip_bfloat16* op = out + (long)b * out_stride + (long)(kh * G + col) * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      const int dim0 = wid * D4 + dt * 16 + kgrp * 4;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        op[dim0 + reg] = __float2bfloat16(ot[dt][reg] * inv);
    }
  }
}

// ------------------------------------------- ring-buffered MFMA decode --
// Variant of paged_decode_pipe_kernel with SEPARATE K and V slot pairs and
// 32-key chunks: both V(i) and K(i+1) are issued at the LOOP TOP, so every
// DMA gets a full iteration to stream (the shared-buffer kernel can only
// issue V after S finishes reading the buffer — V's stream window is just
// the softmax phase). Costs 3 barriers per 32 keys (vs 3 per 64); the bet
// is that removing the V stall beats the barrier overhead. A/B via
// LLMQ_DECODE_PIPE=32.
template <int HEAD_DIM, int NW>
__global__ __launch_bounds__(NW * WAVE, 4) void paged_decode_ring_kernel(
    __hip_bfloat16* __restrict__ out,      // [B, H, D]
    const __hip_bfloat16* __restrict__ q,  // [B, H, D]
    const __hip_bfloat16* __restrict__ k_cache,  // [nb, KVH, bs, D]
    const __hip_bfloat16* __restrict__ v_cache,
    const int* __restrict__ block_tables,  // [B, max_blocks<=PD_MAX_BT]
    const int* __restrict__ context_lens,  // [B]
    float* __restrict__ scratch,           // [B,KVH,NSPLIT,G,D+2] (NSPLIT>1)
    int num_heads, int num_kv_heads, int block_size, int max_blocks,
    float scale, float softcap, int window, long q_stride, long out_stride) {
  constexpr int D = HEAD_DIM;
  constexpr int KS = D / 32;
  constexpr int PD_KT = 32;
  constexpr int NT = NW * WAVE;
  constexpr int SLABS = PD_KT / 16;   // 2
  constexpr int D4 = D / NW;
  constexpr int DT = D4 / 16;
  constexpr int CPK = D / 8;
  constexpr int NI_K = PD_KT * CPK / NT;
  constexpr int NSUB = (PD_KT / 32) * (D / 16);
  constexpr int NI_V = NSUB / NW;
  constexpr int K_ELEMS = PD_KT * D;
  constexpr int V_ELEMS = NSUB * 528;
  static_assert(NI_K >= 1 && NI_V >= 1 && D % NW == 0 && D4 % 16 == 0, "");

  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int G = num_heads / num_kv_heads;
  const int L = context_lens[b];
  if (L <= 0) return;

  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int col = lane & 15;
  const int kgrp = lane >> 4;
  const int slab = wid % SLABS;

  constexpr int P_OFF = 2 * K_ELEMS + 2 * V_ELEMS;      // shorts
  constexpr int F_BASE = (P_OFF + 16 * PD_KT + 7) / 8 * 4;  // float idx
  constexpr int ALPHA_F = F_BASE + SLABS * 16;
  constexpr int LPART_F = ALPHA_F + 16;
  constexpr int BT_I = LPART_F + SLABS * 16;
  constexpr int TOTAL_BYTES = BT_I * 4 + PD_MAX_BT * 4;
  __shared__ __attribute__((aligned(16))) char smem[TOTAL_BYTES];
  short* const ks0 = reinterpret_cast<short*>(smem);
  short* const ks1 = ks0 + K_ELEMS;
  short* const vs0 = ks1 + K_ELEMS;
  short* const vs1 = vs0 + V_ELEMS;
  short* const p_lds2 = ks0 + P_OFF;                    // [16][PD_KT]
  float* const fbase = reinterpret_cast<float*>(smem);
  float* const mpart = fbase + F_BASE;
  float* const alpha_s = fbase + ALPHA_F;
  float* const l_part = fbase + LPART_F;
  int* const bt_l = reinterpret_cast<int*>(smem) + BT_I;

  const int* bt_glob = block_tables + (long)b * max_blocks;
  {
    const int nb = (L + block_size - 1) / block_size;
    for (int i = tid; i < nb; i += NT) bt_l[i] = bt_glob[i];
  }

  bf16x8_t qfrag[KS];
  {
    const int qrow = (col < G) ? col : 0;
    const __hip_bfloat16* qp = q + (long)b * q_stride + (long)(kh * G + qrow) * D;
#pragma unroll
    for (int ks = 0; ks < KS; ++ks) {
      qfrag[ks] = *reinterpret_cast<const bf16x8_t*>(qp + ks * 32 + kgrp * 8);
      reg_fence(qfrag[ks]);
    }
  }

  float m_regs[4] = {-1e30f, -1e30f, -1e30f, -1e30f};
  float l_regs[4] = {0.f, 0.f, 0.f, 0.f};
  f32x4_t ot[DT];
#pragma unroll
  for (int dt = 0; dt < DT; ++dt) ot[dt] = {0.f, 0.f, 0.f, 0.f};

  const int start = (window > 0 && L > window) ? (L - window) : 0;
  const int base0 = (start / PD_KT) * PD_KT;
  const int nsplit = gridDim.z;
  int range_lo = base0, range_hi = L;
  if (nsplit > 1) {
    const int nchunks = (L - base0 + PD_KT - 1) / PD_KT;
    const int per = (nchunks + nsplit - 1) / nsplit;
    range_lo = base0 + (int)blockIdx.z * per * PD_KT;
    range_hi = min(L, range_lo + per * PD_KT);
  }

  auto k_src = [&](int base, int g) -> const __hip_bfloat16* {
    const int key = g / CPK;
    const int r8 = (g % CPK) * 8;
    const int d = r8 ^ ((key & 7) << 3);
    const int gkey = min(base + key, L - 1);
    const long blk = bt_l[gkey / block_size];
    return k_cache + (blk * num_kv_heads + kh) * ((long)block_size * D) +
           (long)(gkey % block_size) * D + d;
  };
  auto v_src = [&](int base, int st) -> const __hip_bfloat16* {
    const int ks = st / (D / 16), dtile = st % (D / 16);
    const int pos = lane * 8;
    const int bpos = pos / 64, rem = pos % 64;
    const int qq = ((bpos >> 2) & 1) | ((bpos & 3) << 1);
    const int key = ks * 32 + qq * 4 + rem / 16;
    const int dim = dtile * 16 + (rem & 15);
    const int gkey = min(base + key, L - 1);
    const long blk = bt_l[gkey / block_size];
    return v_cache + (blk * num_kv_heads + kh) * ((long)block_size * D) +
           (long)(gkey % block_size) * D + dim;
  };
  auto issue_k = [&](int base, short* dst) {
#pragma unroll
    for (int j = 0; j < NI_K; ++j) {
      const int inst = wid * NI_K + j;
      const int off = __builtin_amdgcn_readfirstlane(inst * 512);
      glds16(k_src(base, inst * 64 + lane), dst + off);
    }
  };
  auto issue_v = [&](int base, short* dst) {
#pragma unroll
    for (int j = 0; j < NI_V; ++j) {
      const int st = wid * NI_V + j;
      const int off = __builtin_amdgcn_readfirstlane(st * 528);
      glds16(v_src(base, st), dst + off);
    }
  };

  pipe_barrier();  // bt_l ready
  if (range_lo < range_hi) issue_k(range_lo, ks0);

  int cur = 0;
  for (int base = range_lo; base < range_hi; base += PD_KT, cur ^= 1) {
    short* const KX = cur ? ks1 : ks0;
    short* const KY = cur ? ks0 : ks1;
    short* const VX = cur ? vs1 : vs0;
    // K(base) landed; V slot VX held V(base-2): PV(base-2) finished two
    // barriers ago; KY held K(base-1): S(base-1) done.
    pipe_barrier_vm<0>();
    issue_v(base, VX);
    const int next = base + PD_KT;
    const bool prefetch = next < range_hi;
    if (prefetch) issue_k(next, KY);

    // ---- S[16,16] for this wave's slab
    f32x4_t s = {0.f, 0.f, 0.f, 0.f};
    {
      const int key = slab * 16 + col;
#pragma unroll
      for (int ks = 0; ks < KS; ++ks) {
        const int d8 = ks * 32 + kgrp * 8;
        bf16x8_t bfrag =
            *reinterpret_cast<const bf16x8_t*>(&KX[key * D + swz(key, d8)]);
        s = __builtin_amdgcn_mfma_f32_16x16x32_bf16(qfrag[ks], bfrag, s, 0, 0, 0);
      }
    }
    float sv[4];
    {
      const int key = base + slab * 16 + col;
      const bool dead = key >= L || key < start;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        float x = s[reg] * scale;
        if (softcap > 0.f) x = tanhf(x / softcap) * softcap;
        sv[reg] = dead ? -1e30f : x;
        float m = sv[reg];
#pragma unroll
        for (int off = 1; off < 16; off <<= 1) m = fmaxf(m, __shfl_xor(m, off, WAVE));
        if (col == 0 && wid < SLABS) mpart[slab * 16 + kgrp * 4 + reg] = m;
      }
    }
    pipe_barrier();  // mparts visible (K reads done; no buffer handoff)

#pragma unroll
    for (int reg = 0; reg < 4; ++reg) {
      const int row = kgrp * 4 + reg;
      float m_tile = mpart[row];
#pragma unroll
      for (int w = 1; w < SLABS; ++w) m_tile = fmaxf(m_tile, mpart[w * 16 + row]);
      const float m_new = fmaxf(m_regs[reg], m_tile);
      const float alpha = (m_new > -1e30f) ? __expf(m_regs[reg] - m_new) : 1.f;
      const float pe = (m_new > -1e30f && sv[reg] > -1e29f)
                           ? __expf(sv[reg] - m_new) : 0.f;
      float lsum = pe;
#pragma unroll
      for (int off = 1; off < 16; off <<= 1) lsum += __shfl_xor(lsum, off, WAVE);
      l_regs[reg] = l_regs[reg] * alpha + lsum;
      m_regs[reg] = m_new;
      if (wid == 0 && col == 0) alpha_s[row] = alpha;
      p_lds2[row * PD_KT + slab * 16 + col] =
          __bfloat16_as_short(__float2bfloat16(pe));
    }
    if (prefetch) {
      pipe_barrier_vm<NI_K>();  // V(base) landed; K(next) stays in flight
    } else {
      pipe_barrier_vm<0>();
    }

    const float alpha_q = alpha_s[col];
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      ot[dt][0] *= alpha_q; ot[dt][1] *= alpha_q;
      ot[dt][2] *= alpha_q; ot[dt][3] *= alpha_q;
      const int dtile = (wid * D4) / 16 + dt;
#pragma unroll
      for (int ks = 0; ks < PD_KT / 32; ++ks) {
        const int sub = (ks * (D / 16) + dtile) * 528 + lane * 4;
        bf16x4v lo = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4*)&VX[sub]);
        bf16x4v hi = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
            (lds_bf16x4*)&VX[sub + 4 * 64]);
        bf16x8_t a;
#pragma unroll
        for (int j = 0; j < 4; ++j) {
          a[j] = __bfloat16_as_short((__hip_bfloat16)lo[j]);
          a[4 + j] = __bfloat16_as_short((__hip_bfloat16)hi[j]);
        }
        bf16x8_t bb = *reinterpret_cast<const bf16x8_t*>(
            &p_lds2[col * PD_KT + ks * 32 + kgrp * 8]);
        ot[dt] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, bb, ot[dt], 0, 0, 0);
      }
    }
  }

  // ---- merge per-slab l, then store
  if (col == 0 && wid < SLABS) {
#pragma unroll
    for (int reg = 0; reg < 4; ++reg) l_part[slab * 16 + kgrp * 4 + reg] = l_regs[reg];
  }
  pipe_barrier();
  float l_tot = l_part[col];
#pragma unroll
  for (int w = 1; w < SLABS; ++w) l_tot += l_part[w * 16 + col];
  if (nsplit > 1) {
    float* slot = scratch +
        ((((long)b * num_kv_heads + kh) * nsplit + blockIdx.z) * G) * (D + 2);
    if (col < G) {
      float* row = slot + (long)col * (D + 2);
#pragma unroll
      for (int dt = 0; dt < DT; ++dt) {
        const int dim0 = wid * D4 + dt * 16 + kgrp * 4;
#pragma unroll
        for (int reg = 0; reg < 4; ++reg) row[dim0 + reg] = ot[dt][reg];
      }
    }
    if (wid == 0 && col == 0) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int r = kgrp * 4 + reg;
        if (r < G) {
          float* row = slot + (long)r * (D + 2);
          row[D] = m_regs[reg];
          float lt = l_part[r];
          for (int w = 1; w < SLABS; ++w) lt += l_part[w * 16 + r];
          row[D + 1] = lt;
        }
      }
    }
    return;
  }
  const float inv = (l_tot > 0.f) ? 1.0f / l_tot : 0.f;
  if (col < G) {
    __hip_bfloat16* op = out + (long)b * out_stride + (long)(kh * G + col) * D;
#pragma unroll
    for (int dt = 0; dt < DT; ++dt) {
      const int dim0 = wid * D4 + dt * 16 + kgrp * 4;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        op[dim0 + reg] = __float2bfloat16(ot[dt][reg] * inv);
    }
  }
}

// Combine split-KV partials: out[b, kh*G+g] = Σ_s w_s·acc_s / Σ_s w_s·l_s,
// w_s = exp(m_s − max_s m). One wave per (b, kh, g).
template <int HEAD_DIM>
__global__ __launch_bounds__(64) void decode_splitkv_merge_kernel(
    __hip_bfloat16* __restrict__ out, const float* __restrict__ scratch,
    int num_kv_heads, int G, int nsplit, long out_stride) {
  constexpr int D = HEAD_DIM;
  constexpr int DPL = D / WAVE;  // dims per lane (2 or 4)
  const int b = blockIdx.x;
  const int kh = blockIdx.y;
  const int g = blockIdx.z;
  const int lane = threadIdx.x;
  const float* base = scratch +
      ((((long)b * num_kv_heads + kh) * nsplit) * G + g) * (D + 2);
  const long sstride = (long)G * (D + 2);
  float m_max = -1e30f;
  for (int s2 = 0; s2 < nsplit; ++s2) m_max = fmaxf(m_max, base[s2 * sstride + D]);
  float acc[DPL];
#pragma unroll
  for (int i = 0; i < DPL; ++i) acc[i] = 0.f;
  float l = 0.f;
  for (int s2 = 0; s2 < nsplit; ++s2) {
    const float* row = base + s2 * sstride;
    const float w = __expf(row[D] - m_max);
    if (w > 0.f) {
      l += w * row[D + 1];
#pragma unroll
      for (int i = 0; i < DPL; ++i) acc[i] += w * row[lane * DPL + i];
    }
  }
  const float inv = (l > 0.f) ? 1.0f / l : 0.f;
  __hip_bfloat16* op =
      out + (long)b * out_stride + (long)(kh * G + g) * D + lane * DPL;
#pragma unroll
  for (int i = 0; i < DPL; ++i) op[i] = __float2bfloat16(acc[i] * inv);
}
