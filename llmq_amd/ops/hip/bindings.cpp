// Torch bindings for the llmq-amd CDNA4 kernels.
// Registered under torch.ops.llmq_amd.* ; loaded via torch.ops.load_library.

#include <ATen/ATen.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <torch/library.h>

#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>

// Kernel definitions are included BEFORE this file in the umbrella TU
// (ops.hip) — no forward declarations needed (stale declarations with
// narrower template parameter lists silently generate dead launch stubs).

namespace {

#define CHECK_GPU(x) TORCH_CHECK(x.is_cuda(), #x " must be on GPU")
#define CHECK_LASTDIM(x) \
  TORCH_CHECK(x.stride(-1) == 1, #x " innermost dim must be contiguous")

hipStream_t stream() { return c10::hip::getCurrentHIPStream().stream(); }

template <typename F>
void dispatch_dtype(const at::Tensor& t, const char* name, F&& f) {
  switch (t.scalar_type()) {
    case at::kBFloat16:
      f.template operator()<__hip_bfloat16>();
      break;
    case at::kHalf:
      f.template operator()<_Float16>();
      break;
    case at::kFloat:
      f.template operator()<float>();
      break;
    default:
      TORCH_CHECK(false, name, ": unsupported dtype ", t.scalar_type());
  }
}

// ---------------------------------------------------------------- norms --

void rmsnorm(at::Tensor out, at::Tensor in, at::Tensor weight, double eps,
             double offset) {
  CHECK_GPU(in);
  CHECK_LASTDIM(in);
  const int hidden = in.size(-1);
  const long rows = in.numel() / hidden;
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  dispatch_dtype(in, "rmsnorm", [&]<typename T>() {
    hipLaunchKernelGGL(rmsnorm_kernel<T>, dim3(rows), dim3(256), 0, stream(),
                       reinterpret_cast<T*>(out.data_ptr()),
                       reinterpret_cast<const T*>(in.data_ptr()),
                       reinterpret_cast<const T*>(weight.data_ptr()), hidden,
                       (float)eps, (float)offset);
  });
}

void fused_add_rmsnorm(at::Tensor x, at::Tensor residual, at::Tensor weight,
                       double eps, double offset) {
  CHECK_GPU(x);
  CHECK_LASTDIM(x);
  const int hidden = x.size(-1);
  const long rows = x.numel() / hidden;
  dispatch_dtype(x, "fused_add_rmsnorm", [&]<typename T>() {
    hipLaunchKernelGGL(fused_add_rmsnorm_kernel<T>, dim3(rows), dim3(256), 0,
                       stream(), reinterpret_cast<T*>(x.data_ptr()),
                       reinterpret_cast<T*>(residual.data_ptr()),
                       reinterpret_cast<const T*>(weight.data_ptr()), hidden,
                       (float)eps, (float)offset);
  });
}

// ----------------------------------------------------------- activations --

template <bool GELU>
void act_and_mul(at::Tensor out, at::Tensor in) {
  CHECK_GPU(in);
  CHECK_LASTDIM(in);
  const int d = out.size(-1);
  const long rows = out.numel() / d;
  TORCH_CHECK(in.size(-1) == 2 * d, "in must be [..., 2*d]");
  const long chunks = rows * (d / 8);
  const int grid = std::min<long>((chunks + 255) / 256, 2048);
  dispatch_dtype(in, "act_and_mul", [&]<typename T>() {
    hipLaunchKernelGGL((act_and_mul_kernel<T, GELU>), dim3(grid), dim3(256), 0,
                       stream(), reinterpret_cast<T*>(out.data_ptr()),
                       reinterpret_cast<const T*>(in.data_ptr()), rows, d);
  });
}

void silu_and_mul(at::Tensor out, at::Tensor in) { act_and_mul<false>(out, in); }
void gelu_tanh_and_mul(at::Tensor out, at::Tensor in) { act_and_mul<true>(out, in); }

// ----------------------------------------------------------------- rope --

void rope_inplace(at::Tensor q, at::Tensor k, at::Tensor positions,
                  at::Tensor cos_sin) {
  CHECK_GPU(q);
  CHECK_LASTDIM(q);
  CHECK_LASTDIM(k);
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3, "q/k must be [T, H, D]");
  TORCH_CHECK(cos_sin.scalar_type() == at::kFloat, "cos_sin must be f32");
  const int T_ = q.size(0);
  if (T_ == 0) return;
  const int hq = q.size(1), hk = k.size(1), d = q.size(2);
  TORCH_CHECK(q.stride(1) == d && k.stride(1) == d, "head dim must be packed");
  dispatch_dtype(q, "rope", [&]<typename T>() {
    hipLaunchKernelGGL(rope_kernel<T>, dim3(T_), dim3(256), 0, stream(),
                       reinterpret_cast<T*>(q.data_ptr()),
                       reinterpret_cast<T*>(k.data_ptr()),
                       positions.data_ptr<long>(), cos_sin.data_ptr<float>(),
                       hq, hk, d, q.stride(0), k.stride(0));
  });
}

void norm_add_norm(at::Tensor x, at::Tensor residual, at::Tensor w_post,
                   at::Tensor w_pre, double eps, double offset) {
  CHECK_GPU(x);
  CHECK_LASTDIM(x);
  const int hidden = x.size(-1);
  const long rows = x.numel() / hidden;
  dispatch_dtype(x, "norm_add_norm", [&]<typename T>() {
    hipLaunchKernelGGL(norm_add_norm_kernel<T>, dim3(rows), dim3(256), 0,
                       stream(), reinterpret_cast<T*>(x.data_ptr()),
                       reinterpret_cast<T*>(residual.data_ptr()),
                       reinterpret_cast<const T*>(w_post.data_ptr()),
                       reinterpret_cast<const T*>(w_pre.data_ptr()), hidden,
                       (float)eps, (float)offset);
  });
}

// ------------------------------------------------------------- KV cache --

void reshape_and_cache(at::Tensor key, at::Tensor value, at::Tensor k_cache,
                       at::Tensor v_cache, at::Tensor slot_mapping) {
  CHECK_GPU(key);
  CHECK_LASTDIM(key);
  const int T_ = key.size(0);
  if (T_ == 0) return;
  const int kvh = key.size(1), d = key.size(2);
  const int bs = k_cache.size(2);
  TORCH_CHECK(k_cache.is_contiguous() && v_cache.is_contiguous());
  TORCH_CHECK(slot_mapping.scalar_type() == at::kLong);
  const bool fp8_cache = k_cache.scalar_type() == at::kFloat8_e4m3fn;
  dispatch_dtype(key, "reshape_and_cache", [&]<typename T>() {
    if (fp8_cache) {
      if constexpr (std::is_same_v<T, __hip_bfloat16>) {
        hipLaunchKernelGGL((reshape_and_cache_kernel<T, __hip_fp8_e4m3>),
                           dim3(T_), dim3(128), 0, stream(),
                           reinterpret_cast<const T*>(key.data_ptr()),
                           reinterpret_cast<const T*>(value.data_ptr()),
                           reinterpret_cast<__hip_fp8_e4m3*>(k_cache.data_ptr()),
                           reinterpret_cast<__hip_fp8_e4m3*>(v_cache.data_ptr()),
                           slot_mapping.data_ptr<long>(), kvh, d, bs,
                           key.stride(0), value.stride(0));
      } else {
        TORCH_CHECK(false, "fp8 KV cache requires bf16 activations");
      }
      return;
    }
    hipLaunchKernelGGL(reshape_and_cache_kernel<T>, dim3(T_), dim3(128), 0,
                       stream(), reinterpret_cast<const T*>(key.data_ptr()),
                       reinterpret_cast<const T*>(value.data_ptr()),
                       reinterpret_cast<T*>(k_cache.data_ptr()),
                       reinterpret_cast<T*>(v_cache.data_ptr()),
                       slot_mapping.data_ptr<long>(), kvh, d, bs,
                       key.stride(0), value.stride(0));
  });
}

void rope_and_cache(at::Tensor q, at::Tensor k, at::Tensor value,
                    at::Tensor k_cache, at::Tensor v_cache,
                    at::Tensor positions, at::Tensor cos_sin,
                    at::Tensor slot_mapping) {
  CHECK_GPU(q);
  CHECK_LASTDIM(q);
  CHECK_LASTDIM(k);
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3, "q/k must be [T, H, D]");
  TORCH_CHECK(cos_sin.scalar_type() == at::kFloat);
  TORCH_CHECK(slot_mapping.scalar_type() == at::kLong);
  const int T_ = q.size(0);
  if (T_ == 0) return;
  const int hq = q.size(1), hk = k.size(1), d = q.size(2);
  const int bs = k_cache.size(2);
  TORCH_CHECK(q.stride(1) == d && k.stride(1) == d, "head dim must be packed");
  const bool fp8_cache = k_cache.scalar_type() == at::kFloat8_e4m3fn;
  dispatch_dtype(q, "rope_and_cache", [&]<typename T>() {
    if (fp8_cache) {
      if constexpr (std::is_same_v<T, __hip_bfloat16>) {
        hipLaunchKernelGGL((rope_and_cache_kernel<T, __hip_fp8_e4m3>),
                           dim3(T_), dim3(256), 0, stream(),
                           reinterpret_cast<T*>(q.data_ptr()),
                           reinterpret_cast<T*>(k.data_ptr()),
                           reinterpret_cast<const T*>(value.data_ptr()),
                           reinterpret_cast<__hip_fp8_e4m3*>(k_cache.data_ptr()),
                           reinterpret_cast<__hip_fp8_e4m3*>(v_cache.data_ptr()),
                           positions.data_ptr<long>(), cos_sin.data_ptr<float>(),
                           slot_mapping.data_ptr<long>(), hq, hk, d, bs,
                           q.stride(0), k.stride(0), value.stride(0));
      } else {
        TORCH_CHECK(false, "fp8 KV cache requires bf16 activations");
      }
      return;
    }
    hipLaunchKernelGGL(rope_and_cache_kernel<T>, dim3(T_), dim3(256), 0,
                       stream(), reinterpret_cast<T*>(q.data_ptr()),
                       reinterpret_cast<T*>(k.data_ptr()),
                       reinterpret_cast<const T*>(value.data_ptr()),
                       reinterpret_cast<T*>(k_cache.data_ptr()),
                       reinterpret_cast<T*>(v_cache.data_ptr()),
                       positions.data_ptr<long>(), cos_sin.data_ptr<float>(),
                       slot_mapping.data_ptr<long>(), hq, hk, d, bs,
                       q.stride(0), k.stride(0), value.stride(0));
  });
}

// ------------------------------------------------------------ attention --

template <typename T>
void launch_decode(at::Tensor& out, const at::Tensor& q, const at::Tensor& kc,
                   const at::Tensor& vc, const at::Tensor& bt,
                   const at::Tensor& cl, double scale, double softcap,
                   long window) {
  const int B = q.size(0), H = q.size(1), D = q.size(2);
  const int KVH = kc.size(1);
  const int bs = kc.size(2);
  const int max_blocks = bt.size(1);
  const int G = H / KVH;
  TORCH_CHECK(H % KVH == 0, "num_heads must divide num_kv_heads");
  TORCH_CHECK(bs == 16 || bs == 32, "block_size must be 16 or 32");
  dim3 grid(B, KVH);
  // bf16 + MFMA-supported head dims → matrix-core kernel (G padded to 16).
  // NW = waves per workgroup (keys-per-chunk = 16·NW); 8 halves the
  // barriers per token vs 4 at the same waves/SIMD (A/B-able via
  // LLMQ_DECODE_NW).
  if constexpr (std::is_same_v<T, __hip_bfloat16>) {
    if ((D == 128 || D == 256) && G <= 16) {
      static const int nw_env = [] {
        const char* e = getenv("LLMQ_DECODE_NW");
        return e ? atoi(e) : 8;
      }();
      // Split-KV: when B×KVH under-fills the chip (TP ranks hold few kv
      // heads; small batches), split the context across z-workgroups and
      // merge fp32 partials. Target ≥512 workgroups (2 resident per CU).
      int nsplit = 1;
      while (B * KVH * nsplit < 512 && nsplit < 16) nsplit *= 2;
      at::Tensor scratch;
      float* scratch_ptr = nullptr;
      if (nsplit > 1) {
        scratch = at::empty({(long)B, KVH, nsplit, G, D + 2},
                            q.options().dtype(at::kFloat));
        scratch_ptr = scratch.data_ptr<float>();
      }
      dim3 sgrid(B, KVH, nsplit);
      const bool fp8_cache = kc.scalar_type() == at::kFloat8_e4m3fn;
      // Software-pipelined glds kernel (default): LLMQ_DECODE_PIPE selects
      // the chunk size (64 | 128 keys) or 0 = the plain-staged kernel.
      // Read per call so microbenches can A/B in-process. Preconditions:
      // bf16 cache (glds cannot convert fp8 during the LDS write) and the
      // whole block table staged in LDS (no per-granule LDS-vs-global
      // select — guide §5 trap 4c).
      const char* pe = getenv("LLMQ_DECODE_PIPE");
      const int pipe_kt = pe ? atoi(pe) : 64;
      if (fp8_cache && pipe_kt > 0 && max_blocks <= PD_MAX_BT) {
        // fp8 glds pipeline: raw-fp8 staging + native packed cvt VOPs
        auto lf8 = [&]<int HD>() {
          hipLaunchKernelGGL((paged_decode_pipe_fp8_kernel<HD, 8>), sgrid,
                             dim3(8 * 64), 0, stream(),
                             reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                             reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                             reinterpret_cast<const __hip_fp8_e4m3*>(kc.data_ptr()),
                             reinterpret_cast<const __hip_fp8_e4m3*>(vc.data_ptr()),
                             bt.data_ptr<int>(), cl.data_ptr<int>(), scratch_ptr,
                             H, KVH, bs, max_blocks, (float)scale,
                             (float)softcap, (int)window, q.stride(0),
                             out.stride(0));
          if (nsplit > 1) {
            hipLaunchKernelGGL((decode_splitkv_merge_kernel<HD>),
                               dim3(B, KVH, G), dim3(64), 0, stream(),
                               reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                               scratch_ptr, KVH, G, nsplit, out.stride(0));
          }
        };
        if (D == 128) lf8.template operator()<128>();
        else lf8.template operator()<256>();
        return;
      }
      if (!fp8_cache && pipe_kt > 0 && max_blocks <= PD_MAX_BT) {
        auto lp = [&]<int HD, int KT>() {
          hipLaunchKernelGGL((paged_decode_pipe_kernel<HD, 8, KT>), sgrid,
                             dim3(8 * 64), 0, stream(),
                             reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                             reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                             reinterpret_cast<const __hip_bfloat16*>(kc.data_ptr()),
                             reinterpret_cast<const __hip_bfloat16*>(vc.data_ptr()),
                             bt.data_ptr<int>(), cl.data_ptr<int>(), scratch_ptr,
                             H, KVH, bs, max_blocks, (float)scale,
                             (float)softcap, (int)window, q.stride(0),
                             out.stride(0));
          if (nsplit > 1) {
            hipLaunchKernelGGL((decode_splitkv_merge_kernel<HD>),
                               dim3(B, KVH, G), dim3(64), 0, stream(),
                               reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                               scratch_ptr, KVH, G, nsplit, out.stride(0));
          }
        };
        if (pipe_kt == 32) {  // ring variant: separate K/V slots, 32-key
          auto lr = [&]<int HD>() {
            hipLaunchKernelGGL((paged_decode_ring_kernel<HD, 8>), sgrid,
                               dim3(8 * 64), 0, stream(),
                               reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                               reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                               reinterpret_cast<const __hip_bfloat16*>(kc.data_ptr()),
                               reinterpret_cast<const __hip_bfloat16*>(vc.data_ptr()),
                               bt.data_ptr<int>(), cl.data_ptr<int>(), scratch_ptr,
                               H, KVH, bs, max_blocks, (float)scale,
                               (float)softcap, (int)window, q.stride(0),
                               out.stride(0));
            if (nsplit > 1) {
              hipLaunchKernelGGL((decode_splitkv_merge_kernel<HD>),
                                 dim3(B, KVH, G), dim3(64), 0, stream(),
                                 reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                                 scratch_ptr, KVH, G, nsplit, out.stride(0));
            }
          };
          if (D == 128) lr.template operator()<128>();
          else lr.template operator()<256>();
          return;
        }
        if (D == 128) {
          if (pipe_kt >= 128) lp.template operator()<128, 128>();
          else lp.template operator()<128, 64>();
        } else {
          if (pipe_kt >= 128) lp.template operator()<256, 128>();
          else lp.template operator()<256, 64>();
        }
        return;
      }
      auto lm = [&]<int HD, int NW, typename TC>() {
        hipLaunchKernelGGL((paged_decode_mfma_kernel<HD, NW, TC>), sgrid,
                           dim3(NW * 64), 0, stream(),
                           reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                           reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                           reinterpret_cast<const TC*>(kc.data_ptr()),
                           reinterpret_cast<const TC*>(vc.data_ptr()),
                           bt.data_ptr<int>(), cl.data_ptr<int>(), scratch_ptr,
                           H, KVH, bs, max_blocks, (float)scale,
                           (float)softcap, (int)window, q.stride(0),
                           out.stride(0));
        if (nsplit > 1) {
          hipLaunchKernelGGL((decode_splitkv_merge_kernel<HD>),
                             dim3(B, KVH, G), dim3(64), 0, stream(),
                             reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                             scratch_ptr, KVH, G, nsplit, out.stride(0));
        }
      };
      using BF = __hip_bfloat16;
      using F8 = __hip_fp8_e4m3;
      if (fp8_cache) {  // fp8 KV rides the default NW=8 configuration
        if (D == 128) lm.template operator()<128, 8, F8>();
        else lm.template operator()<256, 8, F8>();
      } else if (D == 128) {
        // NW=16 needs D/NW >= 16 (a full MFMA dim tile per wave)
        if (nw_env == 2) lm.template operator()<128, 2, BF>();
        else if (nw_env == 4) lm.template operator()<128, 4, BF>();
        else lm.template operator()<128, 8, BF>();
      } else {
        if (nw_env == 2) lm.template operator()<256, 2, BF>();
        else if (nw_env == 4) lm.template operator()<256, 4, BF>();
        else if (nw_env == 16) lm.template operator()<256, 16, BF>();
        else lm.template operator()<256, 8, BF>();
      }
      return;
    }
  }
  // The VALU fallback reads the cache as T; an fp8 cache reaching it would
  // be silently reinterpreted as bf16 (garbage, no error).
  TORCH_CHECK(kc.scalar_type() != at::kFloat8_e4m3fn,
              "fp8 KV cache requires the MFMA decode path (bf16 q, head_dim "
              "128/256, GQA group <= 16); got head_dim=", D, " group=", G);
  TORCH_CHECK(G <= 8, "GQA group must be <= 8 for the VALU decode kernel");
  const int subs = 256 / (32 * G);
  const int lds = (G * D + DECODE_CHUNK * G + subs * G * (D + 2)) * sizeof(float) +
                  16 * sizeof(int);
  auto l = [&]<int HD>() {
    hipLaunchKernelGGL((paged_decode_attention_kernel<T, HD>), grid, dim3(256),
                       lds, stream(), reinterpret_cast<T*>(out.data_ptr()),
                       reinterpret_cast<const T*>(q.data_ptr()),
                       reinterpret_cast<const T*>(kc.data_ptr()),
                       reinterpret_cast<const T*>(vc.data_ptr()),
                       bt.data_ptr<int>(), cl.data_ptr<int>(), H, KVH, bs,
                       max_blocks, (float)scale, (float)softcap, (int)window,
                       q.stride(0), out.stride(0));
  };
  switch (D) {
    case 64: l.template operator()<64>(); break;
    case 128: l.template operator()<128>(); break;
    case 256: l.template operator()<256>(); break;
    default: TORCH_CHECK(false, "unsupported head_dim ", D);
  }
}

void paged_decode_attention(at::Tensor out, at::Tensor q, at::Tensor k_cache,
                            at::Tensor v_cache, at::Tensor block_tables,
                            at::Tensor context_lens, double scale,
                            double softcap, long window) {
  CHECK_GPU(q);
  CHECK_LASTDIM(q);
  TORCH_CHECK(block_tables.scalar_type() == at::kInt);
  TORCH_CHECK(context_lens.scalar_type() == at::kInt);
  dispatch_dtype(q, "paged_decode_attention", [&]<typename T>() {
    launch_decode<T>(out, q, k_cache, v_cache, block_tables, context_lens,
                     scale, softcap, window);
  });
}

template <typename T>
void launch_prefill(at::Tensor& out, const at::Tensor& q, const at::Tensor& k,
                    const at::Tensor& v, const at::Tensor& cu,
                    const at::Tensor& cu_k, int B,
                    double scale, double softcap, long window, long max_seqlen) {
  const int H = q.size(1), D = q.size(2);
  const int KVH = k.size(1);
  if constexpr (std::is_same_v<T, __hip_bfloat16>) {
    // MFMA flash path (bf16): grid (seq, head, q-tile). Default: the glds
    // software-pipelined kernel; LLMQ_PREFILL_PIPE=0 selects the plain-
    // staged one (A/B). Read per call for in-process microbenches.
    const int qtiles = (int)((max_seqlen + 63) / 64);
    dim3 fgrid(B, H, std::max(qtiles, 1));
    const char* ppe = getenv("LLMQ_PREFILL_PIPE");
    const bool use_pipe = !ppe || atoi(ppe) != 0;
    auto lf = [&]<int HD>() {
      if (use_pipe) {
        hipLaunchKernelGGL((flash_prefill_pipe_kernel<HD>), fgrid, dim3(256), 0,
                           stream(),
                           reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                           reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                           reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),
                           reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),
                           cu.data_ptr<int>(), cu_k.data_ptr<int>(), H, KVH,
                           (float)scale, (float)softcap, (int)window,
                           q.stride(0), k.stride(0), v.stride(0), out.stride(0));
        return;
      }
      hipLaunchKernelGGL((flash_prefill_bf16_kernel<HD>), fgrid, dim3(256), 0,
                         stream(),
                         reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(q.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(k.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(v.data_ptr()),
                         cu.data_ptr<int>(), cu_k.data_ptr<int>(), H, KVH,
                         (float)scale, (float)softcap, (int)window,
                         q.stride(0), k.stride(0), v.stride(0), out.stride(0));
    };
    switch (D) {
      case 64: lf.template operator()<64>(); return;
      case 128: lf.template operator()<128>(); return;
      case 256: lf.template operator()<256>(); return;
      default: TORCH_CHECK(false, "unsupported head_dim ", D);
    }
  }
  dim3 grid(B, H);
  auto l = [&]<int HD>() {
    hipLaunchKernelGGL((varlen_prefill_attention_kernel<T, HD>), grid,
                       dim3(256), 0, stream(),
                       reinterpret_cast<T*>(out.data_ptr()),
                       reinterpret_cast<const T*>(q.data_ptr()),
                       reinterpret_cast<const T*>(k.data_ptr()),
                       reinterpret_cast<const T*>(v.data_ptr()),
                       cu.data_ptr<int>(), cu_k.data_ptr<int>(), H, KVH,
                       (float)scale, (float)softcap, (int)window,
                       q.stride(0), k.stride(0), v.stride(0), out.stride(0));
  };
  switch (D) {
    case 64: l.template operator()<64>(); break;
    case 128: l.template operator()<128>(); break;
    case 256: l.template operator()<256>(); break;
    default: TORCH_CHECK(false, "unsupported head_dim ", D);
  }
}

void varlen_prefill_attention(at::Tensor out, at::Tensor q, at::Tensor k,
                              at::Tensor v, at::Tensor cu_seqlens,
                              at::Tensor cu_seqlens_k,
                              long max_seqlen, double scale, double softcap,
                              long window) {
  CHECK_GPU(q);
  CHECK_LASTDIM(q);
  TORCH_CHECK(cu_seqlens.scalar_type() == at::kInt);
  TORCH_CHECK(cu_seqlens_k.scalar_type() == at::kInt);
  const int B = cu_seqlens.size(0) - 1;
  dispatch_dtype(q, "varlen_prefill_attention", [&]<typename T>() {
    launch_prefill<T>(out, q, k, v, cu_seqlens, cu_seqlens_k, B, scale,
                      softcap, window, max_seqlen);
  });
}

// ---------------------------------------------------------- skinny GEMM --

void skinny_gemm(at::Tensor out, at::Tensor x, at::Tensor w,
                 c10::optional<at::Tensor> bias, long splitk) {
  CHECK_GPU(x);
  CHECK_LASTDIM(x);
  CHECK_LASTDIM(w);
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && w.scalar_type() == at::kBFloat16);
  TORCH_CHECK(out.scalar_type() == at::kBFloat16 && out.is_contiguous());
  const int M = x.size(0), K = x.size(1), N = w.size(0);
  TORCH_CHECK(w.size(1) == K && out.size(0) == M && out.size(1) == N);
  TORCH_CHECK(K % SKG_BK == 0, "K must be a multiple of 64");
  TORCH_CHECK(N % 4 == 0, "N must be a multiple of 4");
  const __hip_bfloat16* bias_ptr = nullptr;
  if (bias.has_value() && bias->defined()) {
    TORCH_CHECK(bias->scalar_type() == at::kBFloat16 && bias->numel() == N);
    bias_ptr = reinterpret_cast<const __hip_bfloat16*>(bias->data_ptr());
  }
  const int mt = (M + SKG_BM - 1) / SKG_BM;
  const int nt = (N + SKG_BN - 1) / SKG_BN;
  int z = (int)splitk;
  if (z <= 0) {  // auto: fill the chip (>=256 blocks), keep >=8 K-steps/slice
    z = 1;
    while (mt * nt * z < 256 && (K / SKG_BK) / (z * 2) >= 8 && z < 8) z *= 2;
  }
  dim3 grid(mt, nt, z);
  static const int sync_dbg = [] {
    const char* e = getenv("LLMQ_SKG_SYNC");
    return e ? atoi(e) : 0;
  }();
  if (z == 1) {
    if (sync_dbg) {
      hipLaunchKernelGGL((skinny_gemm_kernel<0, 1>), grid, dim3(SKG_NT), 0,
                         stream(), out.data_ptr(),
                         reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                         reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                         bias_ptr, M, N, K, x.stride(0), w.stride(0));
      return;
    }
    hipLaunchKernelGGL((skinny_gemm_kernel<0>), grid, dim3(SKG_NT), 0, stream(),
                       out.data_ptr(),
                       reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                       reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                       bias_ptr, M, N, K, x.stride(0), w.stride(0));
    return;
  }
  at::Tensor slabs = at::empty({z, (long)M * N}, x.options().dtype(at::kFloat));
  hipLaunchKernelGGL((skinny_gemm_kernel<1>), grid, dim3(SKG_NT), 0, stream(),
                     slabs.data_ptr(),
                     reinterpret_cast<const __hip_bfloat16*>(x.data_ptr()),
                     reinterpret_cast<const __hip_bfloat16*>(w.data_ptr()),
                     nullptr, M, N, K, x.stride(0), w.stride(0));
  const long MN = (long)M * N;
  const long blocks = (MN / 4 + 255) / 256;
  hipLaunchKernelGGL(skinny_gemm_reduce_kernel, dim3(blocks), dim3(256), 0,
                     stream(),
                     reinterpret_cast<__hip_bfloat16*>(out.data_ptr()),
                     slabs.data_ptr<float>(), bias_ptr, z, MN, N);
}

// ------------------------------------------------------------- sampling --

void topk_topp_bound(at::Tensor out_bound, at::Tensor logits, at::Tensor temps,
                     at::Tensor top_ps, at::Tensor top_ks) {
  CHECK_GPU(logits);
  TORCH_CHECK(logits.scalar_type() == at::kFloat && logits.dim() == 2);
  TORCH_CHECK(out_bound.scalar_type() == at::kFloat);
  TORCH_CHECK(top_ps.scalar_type() == at::kFloat);
  TORCH_CHECK(top_ks.scalar_type() == at::kLong);
  const int B = logits.size(0), V = logits.size(1);
  hipLaunchKernelGGL(topk_topp_bound_kernel, dim3(B), dim3(256), 0, stream(),
                     out_bound.data_ptr<float>(), logits.data_ptr<float>(),
                     temps.data_ptr<float>(), top_ps.data_ptr<float>(),
                     top_ks.data_ptr<long>(), V, logits.stride(0));
}

void sample_gumbel_argmax(at::Tensor out, at::Tensor keys, at::Tensor logits,
                          at::Tensor temps, at::Tensor req_seeds,
                          at::Tensor req_pos, long seed, long step,
                          c10::optional<at::Tensor> bounds) {
  CHECK_GPU(logits);
  TORCH_CHECK(logits.scalar_type() == at::kFloat && logits.dim() == 2);
  TORCH_CHECK(out.scalar_type() == at::kLong);
  TORCH_CHECK(keys.scalar_type() == at::kLong && keys.numel() == logits.size(0));
  TORCH_CHECK(req_seeds.scalar_type() == at::kInt);
  TORCH_CHECK(req_pos.scalar_type() == at::kInt);
  const int B = logits.size(0), V = logits.size(1);
  const float* bounds_ptr = nullptr;
  if (bounds.has_value() && bounds->defined()) {
    TORCH_CHECK(bounds->scalar_type() == at::kFloat && bounds->numel() == B);
    bounds_ptr = bounds->data_ptr<float>();
  }
  // enough splits to fill the chip at small B
  int nsplit = 1;
  while (B * nsplit < 2048 && nsplit < 64 && (V / nsplit) > 4096) nsplit *= 2;
  hipLaunchKernelGGL(sample_argmax_kernel, dim3(B, nsplit), dim3(256), 0,
                     stream(),
                     reinterpret_cast<unsigned long long*>(keys.data_ptr()),
                     logits.data_ptr<float>(), temps.data_ptr<float>(),
                     reinterpret_cast<const unsigned int*>(req_seeds.data_ptr()),
                     reinterpret_cast<const unsigned int*>(req_pos.data_ptr()),
                     bounds_ptr, V, (unsigned int)seed, (unsigned int)step);
  hipLaunchKernelGGL(unpack_keys_kernel, dim3((B + 255) / 256), dim3(256), 0,
                     stream(), out.data_ptr<long>(),
                     reinterpret_cast<const unsigned long long*>(keys.data_ptr()),
                     B);
}

}  // namespace

TORCH_LIBRARY(llmq_amd, m) {
  m.def("rmsnorm(Tensor(a!) out, Tensor input, Tensor weight, float eps, float offset) -> ()");
  m.def("fused_add_rmsnorm(Tensor(a!) x, Tensor(b!) residual, Tensor weight, float eps, float offset) -> ()");
  m.def("silu_and_mul(Tensor(a!) out, Tensor input) -> ()");
  m.def("gelu_tanh_and_mul(Tensor(a!) out, Tensor input) -> ()");
  m.def("rope_inplace(Tensor(a!) q, Tensor(b!) k, Tensor positions, Tensor cos_sin) -> ()");
  m.def("reshape_and_cache(Tensor key, Tensor value, Tensor(a!) k_cache, Tensor(b!) v_cache, Tensor slot_mapping) -> ()");
  m.def("paged_decode_attention(Tensor(a!) out, Tensor q, Tensor k_cache, Tensor v_cache, Tensor block_tables, Tensor context_lens, float scale, float softcap, int window) -> ()");
  m.def("varlen_prefill_attention(Tensor(a!) out, Tensor q, Tensor k, Tensor v, Tensor cu_seqlens, Tensor cu_seqlens_k, int max_seqlen, float scale, float softcap, int window) -> ()");
  m.def("sample_gumbel_argmax(Tensor(a!) out, Tensor(b!) keys, Tensor logits, Tensor temps, Tensor req_seeds, Tensor req_pos, int seed, int step, Tensor? bounds=None) -> ()");
  m.def("topk_topp_bound(Tensor(a!) out_bound, Tensor logits, Tensor temps, Tensor top_ps, Tensor top_ks) -> ()");
  m.def("skinny_gemm(Tensor(a!) out, Tensor x, Tensor w, Tensor? bias, int splitk) -> ()");
  m.def("norm_add_norm(Tensor(a!) x, Tensor(b!) residual, Tensor w_post, Tensor w_pre, float eps, float offset) -> ()");
  m.def("rope_and_cache(Tensor(a!) q, Tensor(b!) k, Tensor value, Tensor(c!) k_cache, Tensor(d!) v_cache, Tensor positions, Tensor cos_sin, Tensor slot_mapping) -> ()");
}

TORCH_LIBRARY_IMPL(llmq_amd, CUDA, m) {
  m.impl("rmsnorm", &rmsnorm);
  m.impl("fused_add_rmsnorm", &fused_add_rmsnorm);
  m.impl("silu_and_mul", &silu_and_mul);
  m.impl("gelu_tanh_and_mul", &gelu_tanh_and_mul);
  m.impl("rope_inplace", &rope_inplace);
  m.impl("reshape_and_cache", &reshape_and_cache);
  m.impl("paged_decode_attention", &paged_decode_attention);
  m.impl("varlen_prefill_attention", &varlen_prefill_attention);
  m.impl("sample_gumbel_argmax", &sample_gumbel_argmax);
  m.impl("topk_topp_bound", &topk_topp_bound);
  m.impl("skinny_gemm", &skinny_gemm);
  m.impl("norm_add_norm", &norm_add_norm);
  m.impl("rope_and_cache", &rope_and_cache);
}
