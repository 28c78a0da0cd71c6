// Skinny-M bf16 GEMM for the decode projections: out[M,N] = x[M,K] @ W[N,K]^T.
//
// The serving decode step at batch 512 spends ~13.5 ms in hipBLASLt at
// these shapes (M=512, K=3584..14336, N=3584..256k) — ≈700 TF, 4.6× the
// weight-stream floor (profiles/r2_step2). This kernel is the CDNA4
// guide's 256² tile / BK=64 / 8-wave structure (its "8-phase template"):
// both operands are k-contiguous ([M,K] activations, [N,K] torch linear
// weights), staged to LDS by global_load_lds with the st_16x32 XOR swizzle
// on the SOURCE address (lane-linear dest, rule 21), MFMA 16x16x32_bf16,
// counted-vmcnt software pipeline with raw barriers.
//
// Split-K: gridDim.z slices of K accumulate fp32 partial slabs
// [SPLITK, M, N]; a separate reduce kernel folds them (+ optional bias) to
// bf16. SPLITK == 1 writes bf16 directly (bias folded in the epilogue).
//
// Edge handling: M and N tails clamp the SOURCE row (finite real data) and
// out-of-range outputs are simply not stored — no zero-fill staging, the
// pipeline shape never changes.

#include "common.h"

#ifndef SKG_TYPES
#define SKG_TYPES
typedef __attribute__((ext_vector_type(8))) short skg_bf16x8;
typedef __attribute__((ext_vector_type(4))) float skg_f32x4;
#endif

#define SKG_BM 256
#define SKG_BN 256
#define SKG_BK 64
#define SKG_NT 512  // 8 waves: 2 (M) x 4 (N)

// LDS: double-buffered [BM][BK] A-tile + [BN][BK] B-tile, bf16, linear
// layout (glds writes lane-linear), st_16x32 swizzle carried by the
// source/read addresses: elem col ^= ((row>>3)&1)<<4.
#define SKG_TILE_ELEMS (SKG_BM * SKG_BK)

DEVINL int skg_swz(int row, int col) { return col ^ (((row >> 3) & 1) << 4); }

// Stage one [256][64] bf16 tile (A or B) via glds: 2048 granules of 16B
// (32 KB), 4 instructions per wave (8 waves x 64 lanes x 4). Source row
// clamped to the tensor (finite garbage beyond the tail; outputs there
// are never stored).
DEVINL void skg_stage(const __hip_bfloat16* __restrict__ src, long src_stride,
                      int row0, int rows_total, int k0, short* dst,
                      int wid, int lane) {
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const int g = (wid * 4 + j) * 64 + lane;   // granule 0..2047
    const int row = g >> 3;                    // 8 granules per 64-elem row
    const int col = (g & 7) * 8;
    const int srow = min(row0 + row, rows_total - 1);
    const int scol = k0 + skg_swz(row, col);
    const int off = __builtin_amdgcn_readfirstlane((wid * 4 + j) * 512);
    glds16(src + (long)srow * src_stride + scol, dst + off);
  }
}

// SYNC_DEBUG: plain-load + ds_write staging with __syncthreads (no glds,
// no pipeline) — used to isolate staging-pipeline bugs from layout bugs.
template <int SPLITK_TAG, int SYNC_DEBUG = 0>
__global__ __launch_bounds__(SKG_NT, 2) void skinny_gemm_kernel(
    void* __restrict__ out_raw,            // bf16 [M,N] or f32 [Z,M,N]
    const __hip_bfloat16* __restrict__ x,  // [M,K]
    const __hip_bfloat16* __restrict__ w,  // [N,K]
    const __hip_bfloat16* __restrict__ bias,  // [N] or null (SPLITK==1 only)
    int M, int N, int K, long x_stride, long w_stride) {
  const int mtile = blockIdx.x;            // M / 256 tiles
  const int ntile = blockIdx.y;
  const int nsplit = gridDim.z;
  const int kslice = blockIdx.z;
  // K range of this slice (64-aligned split)
  const int ksteps_total = K / SKG_BK;
  const int per = (ksteps_total + nsplit - 1) / nsplit;
  const int ks0 = kslice * per;
  const int ks1 = min(ksteps_total, ks0 + per);
  if (ks0 >= ks1) return;  // (never for supported shapes; guard anyway)

  const int tid = threadIdx.x;
  const int wid = tid / WAVE;
  const int lane = tid & (WAVE - 1);
  const int wm = wid >> 2;       // 0..1  (M half)
  const int wn = wid & 3;        // 0..3  (N quarter: 64 cols)
  const int col16 = lane & 15;
  const int kgrp = lane >> 4;

  __shared__ __attribute__((aligned(16))) short smem[4 * SKG_TILE_ELEMS];
  short* const a0 = smem;
  short* const b0 = smem + SKG_TILE_ELEMS;
  short* const a1 = smem + 2 * SKG_TILE_ELEMS;
  short* const b1 = smem + 3 * SKG_TILE_ELEMS;

  const int row_a0 = mtile * SKG_BM;
  const int row_b0 = ntile * SKG_BN;

  // accumulators: per-wave 128(M) x 64(N) = 8 x 4 fragments of 16x16
  skg_f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  auto stage_sync = [&](const __hip_bfloat16* src, long stride, int row0,
                        int rows_total, int k0, short* dst) {
    for (int g = tid * 2; g < 2048; g += SKG_NT * 2) {
#pragma unroll
      for (int j = 0; j < 2; ++j) {
        const int gg = g + j;
        const int row = gg >> 3;
        const int col = (gg & 7) * 8;
        const int srow = min(row0 + row, rows_total - 1);
        const int scol = k0 + skg_swz(row, col);
        *reinterpret_cast<skg_bf16x8*>(&dst[gg * 8]) =
            *reinterpret_cast<const skg_bf16x8*>(src + (long)srow * stride + scol);
      }
    }
  };

  // prologue: stage K-step ks0 into buffer 0
  if constexpr (!SYNC_DEBUG) {
    skg_stage(x, x_stride, row_a0, M, ks0 * SKG_BK, a0, wid, lane);
    skg_stage(w, w_stride, row_b0, N, ks0 * SKG_BK, b0, wid, lane);
  }

  for (int ks = ks0; ks < ks1; ++ks) {
    short* const a = (ks - ks0) & 1 ? a1 : a0;
    short* const b = (ks - ks0) & 1 ? b1 : b0;
    short* const an = (ks - ks0) & 1 ? a0 : a1;
    short* const bn = (ks - ks0) & 1 ? b0 : b1;
    if constexpr (SYNC_DEBUG) {
      __syncthreads();
      stage_sync(x, x_stride, row_a0, M, ks * SKG_BK, a);
      stage_sync(w, w_stride, row_b0, N, ks * SKG_BK, b);
      __syncthreads();
    } else {
      // current tile landed (the 4 staging glds of this buffer)
      pipe_barrier_vm<0>();
      // prefetch next K-step into the other buffer (stays in flight
      // through the MFMA phase; drained by the next iteration's barrier)
      if (ks + 1 < ks1) {
        skg_stage(x, x_stride, row_a0, M, (ks + 1) * SKG_BK, an, wid, lane);
        skg_stage(w, w_stride, row_b0, N, (ks + 1) * SKG_BK, bn, wid, lane);
      }
    }
    // MFMA over this K-step: per wave 8 m-frags x 4 n-frags x 2 k-substeps
#pragma unroll
    for (int kk = 0; kk < 2; ++kk) {  // K substep of 32
      // B fragments for this wave's 4 n-frags
      skg_bf16x8 bf[4];
#pragma unroll
      for (int nf = 0; nf < 4; ++nf) {
        const int brow = wn * 64 + nf * 16 + col16;
        const int bcol = kk * 32 + kgrp * 8;
        bf[nf] = *reinterpret_cast<const skg_bf16x8*>(
            &b[brow * SKG_BK + skg_swz(brow, bcol)]);
      }
#pragma unroll
      for (int mf = 0; mf < 8; ++mf) {
        const int arow = wm * 128 + mf * 16 + col16;
        const int acol = kk * 32 + kgrp * 8;
        const skg_bf16x8 af = *reinterpret_cast<const skg_bf16x8*>(
            &a[arow * SKG_BK + skg_swz(arow, acol)]);
#pragma unroll
        for (int nf = 0; nf < 4; ++nf) {
          acc[mf][nf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              af, bf[nf], acc[mf][nf], 0, 0, 0);
        }
      }
    }
    // No trailing barrier: the next iteration's vm<0> barrier both drains
    // the prefetch and (being a barrier) guarantees every wave's LDS reads
    // of this buffer finished before anyone stages over the other one.
  }

  // ---- epilogue: C fragment (row = kgrp*4+reg, col = col16)
#pragma unroll
  for (int mf = 0; mf < 8; ++mf) {
    const int gm_base = mtile * SKG_BM + wm * 128 + mf * 16 + kgrp * 4;
#pragma unroll
    for (int nf = 0; nf < 4; ++nf) {
      const int gn = ntile * SKG_BN + wn * 64 + nf * 16 + col16;
      if (gn >= N) continue;
#pragma unroll
      for (int reg = 0; reg < 4; ++reg) {
        const int gm = gm_base + reg;
        if (gm >= M) continue;
        if constexpr (SPLITK_TAG == 0) {
          float v = acc[mf][nf][reg];
          if (bias != nullptr) v += __bfloat162float(bias[gn]);
          reinterpret_cast<__hip_bfloat16*>(out_raw)[(long)gm * N + gn] =
              __float2bfloat16(v);
        } else {
          reinterpret_cast<float*>(out_raw)[((long)kslice * M + gm) * N + gn] =
              acc[mf][nf][reg];
        }
      }
    }
  }
}

// Fold split-K fp32 slabs [Z,M,N] (+ bias) -> bf16 [M,N].
__global__ __launch_bounds__(256) void skinny_gemm_reduce_kernel(
    __hip_bfloat16* __restrict__ out, const float* __restrict__ slabs,
    const __hip_bfloat16* __restrict__ bias, int nsplit, long MN, int N) {
  const long i0 = ((long)blockIdx.x * 256 + threadIdx.x) * 4;
  if (i0 >= MN) return;
  float v[4] = {0.f, 0.f, 0.f, 0.f};
  for (int s = 0; s < nsplit; ++s) {
    const float4 sl = *reinterpret_cast<const float4*>(&slabs[s * MN + i0]);
    v[0] += sl.x; v[1] += sl.y; v[2] += sl.z; v[3] += sl.w;
  }
#pragma unroll
  for (int j = 0; j < 4; ++j) {
    const long i = i0 + j;
    float r = v[j];
    if (bias != nullptr) r += __bfloat162float(bias[i % N]);
    out[i] = __float2bfloat16(r);
  }
}
