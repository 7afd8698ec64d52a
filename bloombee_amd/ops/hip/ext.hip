// Torch extension bindings for the bloombee_amd gfx950 kernel library.
// Host-side shape checks, template dispatch, and stream plumbing only —
// all compute lives in the .hip translation units.

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>
#include <cstdlib>

#include "common.h"

// Single translation unit: the template kernels must be visible at their
// launch sites for instantiation.
#include "elementwise.hip"
#include "kvcache.hip"
#include "attn_decode.hip"
#include "attn_decode_mfma.hip"
#include "attn_prefill.hip"
#include "gemm_skinny.hip"
#include "moe_gemm.hip"
#include "w4_gemm.hip"
#include "mfma_selftest.hip"
#include "quant4.hip"
#include "wire.h"

typedef __attribute__((ext_vector_type(8))) short short8_h;

#define CHECK_DEV(x) TORCH_CHECK(x.is_cuda(), #x " must be on device")
// a CPU tensor pointer reaching a kernel is a GPU memory fault at a random
// later point — check EVERY tensor argument, not just the big ones
#define CHECK_DEV_ALL1(a) CHECK_DEV(a)
#define CHECK_DEV_ALL2(a, b) { CHECK_DEV(a); CHECK_DEV(b); }
#define CHECK_DEV_ALL3(a, b, c) { CHECK_DEV(a); CHECK_DEV(b); CHECK_DEV(c); }
#define CHECK_BF16(x) TORCH_CHECK(x.scalar_type() == at::kBFloat16, #x " must be bf16")
#define CHECK_CONTIG(x) TORCH_CHECK(x.is_contiguous(), #x " must be contiguous")

static inline const unsigned short* bf_ptr(const torch::Tensor& t) {
  return reinterpret_cast<const unsigned short*>(t.data_ptr());
}
static inline unsigned short* bf_ptr_mut(torch::Tensor& t) {
  return reinterpret_cast<unsigned short*>(t.data_ptr());
}
static inline hipStream_t cur_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

// ---------------------------------------------------------------------------
// norms
// ---------------------------------------------------------------------------

template <typename F>
static void dispatch_iters(int H, F f) {
  constexpr int BLOCK = 256;
  const int per_iter = BLOCK * 8;
  if (H <= per_iter) f(std::integral_constant<int, 1>{});
  else if (H <= 2 * per_iter) f(std::integral_constant<int, 2>{});
  else if (H <= 4 * per_iter) f(std::integral_constant<int, 4>{});
  else if (H <= 8 * per_iter) f(std::integral_constant<int, 8>{});
  else TORCH_CHECK(false, "hidden size too large for norm kernel: ", H);
}

torch::Tensor rms_norm(torch::Tensor x, torch::Tensor w, double eps) {
  CHECK_DEV(x); CHECK_BF16(x); CHECK_CONTIG(x);
  const int H = x.size(-1);
  TORCH_CHECK(H % 8 == 0, "H must be a multiple of 8");
  const long N = x.numel() / H;
  auto y = torch::empty_like(x);
  dispatch_iters(H, [&](auto iters) {
    rmsnorm_kernel<256, decltype(iters)::value><<<N, 256, 0, cur_stream()>>>(
        bf_ptr(x), nullptr, bf_ptr(w), nullptr, bf_ptr_mut(y), H, (float)eps);
  });
  return y;
}

std::vector<torch::Tensor> rms_norm_residual(torch::Tensor x, torch::Tensor res,
                                             torch::Tensor w, double eps) {
  CHECK_DEV(x); CHECK_BF16(x); CHECK_CONTIG(x); CHECK_CONTIG(res);
  const int H = x.size(-1);
  const long N = x.numel() / H;
  auto h = torch::empty_like(x);
  auto y = torch::empty_like(x);
  dispatch_iters(H, [&](auto iters) {
    rmsnorm_kernel<256, decltype(iters)::value><<<N, 256, 0, cur_stream()>>>(
        bf_ptr(x), bf_ptr(res), bf_ptr(w), bf_ptr_mut(h), bf_ptr_mut(y), H,
        (float)eps);
  });
  return {h, y};
}

torch::Tensor layer_norm(torch::Tensor x, torch::Tensor w,
                         c10::optional<torch::Tensor> bias, double eps) {
  CHECK_DEV(x); CHECK_BF16(x); CHECK_CONTIG(x);
  const int H = x.size(-1);
  const long N = x.numel() / H;
  auto y = torch::empty_like(x);
  const unsigned short* bp = bias.has_value() ? bf_ptr(*bias) : nullptr;
  dispatch_iters(H, [&](auto iters) {
    layernorm_kernel<256, decltype(iters)::value><<<N, 256, 0, cur_stream()>>>(
        bf_ptr(x), nullptr, bf_ptr(w), bp, nullptr, bf_ptr_mut(y), H, (float)eps);
  });
  return y;
}

// ---------------------------------------------------------------------------
// rope / activations
// ---------------------------------------------------------------------------

void rope_apply_(torch::Tensor q, torch::Tensor k, torch::Tensor cos_t,
                 torch::Tensor sin_t, torch::Tensor pos) {
  CHECK_DEV(q); CHECK_BF16(q); CHECK_CONTIG(q); CHECK_CONTIG(k);
  CHECK_DEV_ALL3(cos_t, sin_t, pos);
  TORCH_CHECK(pos.scalar_type() == at::kInt, "pos must be int32");
  const int B = q.size(0), Hq = q.size(1), T = q.size(2), D = q.size(3);
  const int Hkv = k.size(1);
  TORCH_CHECK(cos_t.scalar_type() == at::kFloat && cos_t.size(1) == D / 2);
  dim3 grid(B * T, Hq + Hkv);
  rope_kernel<<<grid, 64, 0, cur_stream()>>>(
      bf_ptr_mut(q), bf_ptr_mut(k), cos_t.data_ptr<float>(),
      sin_t.data_ptr<float>(), pos.data_ptr<int>(), B, Hq, Hkv, T, D,
      (int)cos_t.size(0));
}

torch::Tensor swiglu(torch::Tensor gu) {
  CHECK_DEV(gu); CHECK_BF16(gu); CHECK_CONTIG(gu);
  const long I = gu.size(-1) / 2;
  const long N = gu.numel() / (2 * I);
  TORCH_CHECK(I % 8 == 0);
  auto sizes = gu.sizes().vec();
  sizes.back() = I;
  auto out = torch::empty(sizes, gu.options());
  const long nvec = N * (I / 8);
  const int grid = (int)std::min<long>((nvec + 255) / 256, 2048);
  swiglu_kernel<<<grid, 256, 0, cur_stream()>>>(bf_ptr(gu), bf_ptr_mut(out), N, I);
  return out;
}

torch::Tensor gelu_tanh(torch::Tensor x) {
  CHECK_DEV(x); CHECK_BF16(x); CHECK_CONTIG(x);
  auto out = torch::empty_like(x);
  const long nvec = x.numel() / 8;
  const int grid = (int)std::min<long>((nvec + 255) / 256, 2048);
  gelu_kernel<<<grid, 256, 0, cur_stream()>>>(bf_ptr(x), bf_ptr_mut(out), x.numel());
  return out;
}

torch::Tensor add_bf16(torch::Tensor a, torch::Tensor b) {
  CHECK_DEV(a); CHECK_BF16(a); CHECK_CONTIG(a); CHECK_CONTIG(b);
  auto out = torch::empty_like(a);
  const long nvec = a.numel() / 8;
  const int grid = (int)std::min<long>((nvec + 255) / 256, 2048);
  add_kernel<<<grid, 256, 0, cur_stream()>>>(bf_ptr(a), bf_ptr(b), bf_ptr_mut(out),
                                             a.numel());
  return out;
}

// ---------------------------------------------------------------------------
// paged KV
// ---------------------------------------------------------------------------

void kv_write(torch::Tensor k_new, torch::Tensor v_new, torch::Tensor k_pages,
              torch::Tensor v_pages, torch::Tensor page_table,
              torch::Tensor start_pos) {
  CHECK_DEV(k_new); CHECK_BF16(k_new); CHECK_CONTIG(k_new); CHECK_CONTIG(v_new);
  CHECK_CONTIG(k_pages); CHECK_CONTIG(v_pages);
  TORCH_CHECK(page_table.scalar_type() == at::kInt && start_pos.scalar_type() == at::kInt);
  CHECK_DEV_ALL2(page_table, start_pos);
  const int B = k_new.size(0), Hkv = k_new.size(1), T = k_new.size(2), D = k_new.size(3);
  const int P = k_pages.size(2), maxp = page_table.size(1);
  TORCH_CHECK(k_pages.size(1) == Hkv && k_pages.size(3) == D);
  TORCH_CHECK(v_pages.size(2) == D && v_pages.size(3) == P,
              "v_pages must be d-major (np, Hkv, D, P)");
  const int threads = std::min(256, Hkv * D / 8);
  kv_write_kernel<<<B * T, threads, 0, cur_stream()>>>(
      bf_ptr(k_new), bf_ptr(v_new), bf_ptr_mut(k_pages), bf_ptr_mut(v_pages),
      page_table.data_ptr<int>(), start_pos.data_ptr<int>(), B, Hkv, T, D, P, maxp);
}

std::vector<torch::Tensor> kv_gather(torch::Tensor k_pages, torch::Tensor v_pages,
                                     torch::Tensor page_table, long batch_index,
                                     long ctx) {
  CHECK_DEV(k_pages); CHECK_DEV(page_table);
  const int Hkv = k_pages.size(1), P = k_pages.size(2), D = k_pages.size(3);
  auto k = torch::empty({Hkv, ctx, D}, k_pages.options());
  auto v = torch::empty({Hkv, ctx, D}, v_pages.options());
  const long nvec = ctx * Hkv * (D / 8);
  const int grid = (int)std::min<long>((nvec + 255) / 256, 2048);
  kv_gather_kernel<<<grid, 256, 0, cur_stream()>>>(
      bf_ptr(k_pages), bf_ptr(v_pages), bf_ptr_mut(k), bf_ptr_mut(v),
      page_table.data_ptr<int>(), (int)batch_index, (int)ctx, Hkv, D, P,
      page_table.size(1));
  return {k, v};
}

// ---------------------------------------------------------------------------
// attention
// ---------------------------------------------------------------------------

template <int D>
static void attn_decode_launch(const torch::Tensor& q, const torch::Tensor& kp,
                               const torch::Tensor& vp, const torch::Tensor& pt,
                               const torch::Tensor& ctx,
                               const float* alibi_ptr, torch::Tensor& out,
                               torch::Tensor& pml, torch::Tensor& pacc, int B,
                               int Hkv, int G, int nch, int P, int maxp,
                               int n_split, int window, float scale, long q_off,
                               long q_sb, long q_sh, long out_sb, long out_sh) {
  dim3 grid(B * Hkv * nch, n_split);
  const int maxg = (G <= 4 && nch == 1) ? 4 : 16;
  // paired-tile variant (NT=2): 2x KV bytes in flight per wave — the
  // latency-bound fix (profiles/r01_st_attention.md SQ_WAIT diagnosis).
  // BBAMD_ATTN_NT=1 reverts; D>=256 keeps NT=1 (VGPR budget).
  static const int nt_env = [] {
    const char* e = std::getenv("BBAMD_ATTN_NT");
    return e ? std::atoi(e) : 2;
  }();
  // Software-pipelined variant (next tile's K+V double-buffered in
  // registers; r02 WAIT_ANY diagnosis): DEFAULT — 50 us / 5.4 TB/s at B32
  // ctx2048 and 6.3 TB/s (achievable-HBM ceiling) at ctx8192 vs 59-62 us
  // for the NT variants. BBAMD_ATTN_PF=0 reverts to the NT path.
  static const bool pf_env = [] {
    const char* e = std::getenv("BBAMD_ATTN_PF");
    return !e || std::atoi(e) != 0;
  }();
  const bool pf = pf_env && (D <= 128);
  const bool nt2 = (nt_env >= 2) && (D <= 128) && !pf;
  auto launch_mfma = [&](auto mg) {
    if constexpr (D == 512) {
      // wide-head path: d split across the 4 waves (gemma-4 global layers)
      attn_decode_wide_kernel<decltype(mg)::value>
          <<<grid, 256, 0, cur_stream()>>>(
              bf_ptr(q) + q_off, bf_ptr(kp), bf_ptr(vp), pt.data_ptr<int>(),
              ctx.data_ptr<int>(), alibi_ptr, bf_ptr_mut(out),
              pml.data_ptr<float>(), pacc.data_ptr<float>(), B, Hkv, G, nch, P,
              maxp, n_split, window, scale, q_sb, q_sh, out_sb, out_sh);
    } else if (D <= 128 && pf) {
      attn_decode_mfma_kernel<D, decltype(mg)::value, 1, 1>
          <<<grid, 256, 0, cur_stream()>>>(
              bf_ptr(q) + q_off, bf_ptr(kp), bf_ptr(vp), pt.data_ptr<int>(),
              ctx.data_ptr<int>(), alibi_ptr, bf_ptr_mut(out),
              pml.data_ptr<float>(), pacc.data_ptr<float>(), B, Hkv, G, nch, P,
              maxp, n_split, window, scale, q_sb, q_sh, out_sb, out_sh);
    } else if (D <= 128 && nt2) {
      attn_decode_mfma_kernel<D, decltype(mg)::value, 2>
          <<<grid, 256, 0, cur_stream()>>>(
              bf_ptr(q) + q_off, bf_ptr(kp), bf_ptr(vp), pt.data_ptr<int>(),
              ctx.data_ptr<int>(), alibi_ptr, bf_ptr_mut(out),
              pml.data_ptr<float>(), pacc.data_ptr<float>(), B, Hkv, G, nch, P,
              maxp, n_split, window, scale, q_sb, q_sh, out_sb, out_sh);
    } else {
      attn_decode_mfma_kernel<D, decltype(mg)::value, 1>
          <<<grid, 256, 0, cur_stream()>>>(
              bf_ptr(q) + q_off, bf_ptr(kp), bf_ptr(vp), pt.data_ptr<int>(),
              ctx.data_ptr<int>(), alibi_ptr, bf_ptr_mut(out),
              pml.data_ptr<float>(), pacc.data_ptr<float>(), B, Hkv, G, nch, P,
              maxp, n_split, window, scale, q_sb, q_sh, out_sb, out_sh);
    }
  };
  if (maxg == 4) launch_mfma(std::integral_constant<int, 4>{});
  else launch_mfma(std::integral_constant<int, 16>{});
  if (n_split > 1) {
    attn_decode_combine_kernel<D>
        <<<B * Hkv * nch, maxg * 16, 0, cur_stream()>>>(
            pml.data_ptr<float>(), pacc.data_ptr<float>(), bf_ptr_mut(out),
            Hkv, G, nch, maxg, n_split, out_sb, out_sh);
  }
}

static torch::Tensor attn_decode_core(
    const torch::Tensor& q, const torch::Tensor& k_pages,
    const torch::Tensor& v_pages, const torch::Tensor& page_table,
    const torch::Tensor& ctx_lens, const c10::optional<torch::Tensor>& alibi,
    double scale, long window, long n_split_req,
    int B, int Hq, int D, long q_off, long q_sb, long q_sh,
    torch::Tensor out, long out_sb, long out_sh) {
  const int Hkv = k_pages.size(1), P = k_pages.size(2), maxp = page_table.size(1);
  const int G = Hq / Hkv;
  TORCH_CHECK(Hq % Hkv == 0);
  TORCH_CHECK(D != 256 || G <= 4, "D=256 decode supports GQA group size <= 4");
  const int nch = (G + 15) / 16;  // MQA chunks (falcon G=71 -> 5)
  const float* alibi_ptr = nullptr;
  if (alibi.has_value()) {
    TORCH_CHECK(alibi->scalar_type() == at::kFloat && alibi->numel() == Hq);
    alibi_ptr = alibi->data_ptr<float>();
  }
  int n_split = (int)n_split_req;
  if (n_split <= 0) {
    // measured optimum (benchmarks/bench_kernels.py attn_decode): size the
    // grid for full residency — NT=1 runs 118 VGPRs -> 4 WGs/CU (1024 WGs),
    // the paired-tile NT=2 variant 158 VGPRs -> 3 WGs/CU (768 WGs); more
    // splits only add merge traffic
    static const int nt_env0 = [] {
      const char* e = std::getenv("BBAMD_ATTN_NT");
      return e ? std::atoi(e) : 2;
    }();
    static const bool pf_env0 = [] {
      const char* e = std::getenv("BBAMD_ATTN_PF");
      return !e || std::atoi(e) != 0;
    }();
    // pipelined variant: 254 VGPRs -> 2 waves/SIMD -> 512 WGs fill the chip
    // (measured: B32 ns=2 50 us / 5.4 TB/s vs 59 at 768-1024 WGs)
    const long target = (pf_env0 && D <= 128) ? 512
                        : (nt_env0 >= 2 && D <= 128) ? 768 : 1024;
    n_split = (int)std::max<long>(
        1, std::min<long>(32, target / std::max(1, B * Hkv * nch)));
  }
  const int maxg = (G <= 4 && nch == 1) ? 4 : 16;
  auto fopt = torch::TensorOptions().device(q.device()).dtype(at::kFloat);
  torch::Tensor pml, pacc;
  if (n_split > 1) {
    pml = torch::empty({(long)B * Hkv * nch * n_split, maxg, 2}, fopt);
    pacc = torch::empty({(long)B * Hkv * nch * n_split, maxg, D}, fopt);
  } else {
    pml = torch::empty({1}, fopt);
    pacc = torch::empty({1}, fopt);
  }
  auto go = [&](auto d) {
    attn_decode_launch<decltype(d)::value>(
        q, k_pages, v_pages, page_table, ctx_lens, alibi_ptr, out, pml, pacc,
        B, Hkv, G, nch, P, maxp, n_split, (int)window, (float)scale, q_off,
        q_sb, q_sh, out_sb, out_sh);
  };
  if (D == 128) go(std::integral_constant<int, 128>{});
  else if (D == 64) go(std::integral_constant<int, 64>{});
  else if (D == 256) go(std::integral_constant<int, 256>{});
  else if (D == 512) go(std::integral_constant<int, 512>{});
  else if (D == 32) go(std::integral_constant<int, 32>{});
  else TORCH_CHECK(false, "unsupported head_dim ", D);
  return out;
}

torch::Tensor attn_decode(torch::Tensor q, torch::Tensor k_pages,
                          torch::Tensor v_pages, torch::Tensor page_table,
                          torch::Tensor ctx_lens, double scale, long window,
                          long n_split_req,
                          c10::optional<torch::Tensor> alibi) {
  CHECK_DEV(q); CHECK_BF16(q); CHECK_CONTIG(q);
  CHECK_DEV_ALL3(k_pages, page_table, ctx_lens);
  if (alibi.has_value()) CHECK_DEV((*alibi));
  TORCH_CHECK(q.dim() == 4 && q.size(2) == 1, "attn_decode expects (B, Hq, 1, D)");
  const int B = q.size(0), Hq = q.size(1), D = q.size(3);
  auto out = torch::empty({B, Hq, 1, D}, q.options());
  return attn_decode_core(q, k_pages, v_pages, page_table, ctx_lens, alibi,
                          scale, window, n_split_req, B, Hq, D,
                          /*q_off*/ 0, (long)Hq * D, (long)D,
                          out, (long)Hq * D, (long)D);
}

// Fused-QKV decode: q section of qkv (B, 1, (Hq+2Hkv)*D) read in place,
// out (B, 1, Hq*D) ready for the O-projection GEMM — zero transposes.
torch::Tensor attn_decode_qkv(torch::Tensor qkv, long Hq, torch::Tensor k_pages,
                              torch::Tensor v_pages, torch::Tensor page_table,
                              torch::Tensor ctx_lens, double scale, long window,
                              long n_split_req) {
  CHECK_DEV(qkv); CHECK_BF16(qkv); CHECK_CONTIG(qkv);
  CHECK_DEV_ALL3(k_pages, page_table, ctx_lens);
  const int Hkv = k_pages.size(1);
  const int D = k_pages.size(3);
  const int B = qkv.size(0);
  TORCH_CHECK(qkv.dim() == 3 && qkv.size(1) == 1 &&
              qkv.size(2) == (Hq + 2 * Hkv) * D, "bad fused qkv shape");
  auto out = torch::empty({B, 1, Hq * D}, qkv.options());
  return attn_decode_core(qkv, k_pages, v_pages, page_table, ctx_lens,
                          c10::nullopt, scale, window, n_split_req, B, (int)Hq,
                          D, /*q_off*/ 0, (long)(Hq + 2 * Hkv) * D, (long)D,
                          out, (long)Hq * D, (long)D)
      .view({B, 1, Hq * D});
}

torch::Tensor attn_prefill(torch::Tensor q, torch::Tensor k_pages,
                           torch::Tensor v_pages, torch::Tensor page_table,
                           torch::Tensor q_start, double scale, long window,
                           c10::optional<torch::Tensor> alibi,
                           c10::optional<torch::Tensor> tree_mask) {
  CHECK_DEV(q); CHECK_BF16(q); CHECK_CONTIG(q);
  CHECK_DEV_ALL3(k_pages, page_table, q_start);
  if (alibi.has_value()) CHECK_DEV((*alibi));
  TORCH_CHECK(q.dim() == 4, "attn_prefill expects (B, Hq, Tq, D)");
  const int B = q.size(0), Hq = q.size(1), Tq = q.size(2), D = q.size(3);
  const int Hkv = k_pages.size(1), P = k_pages.size(2), maxp = page_table.size(1);
  const int G = Hq / Hkv;
  const float* alibi_ptr = nullptr;
  if (alibi.has_value()) {
    TORCH_CHECK(alibi->scalar_type() == at::kFloat && alibi->numel() == Hq);
    alibi_ptr = alibi->data_ptr<float>();
  }
  const unsigned char* tm_ptr = nullptr;
  if (tree_mask.has_value()) {
    TORCH_CHECK(tree_mask->scalar_type() == at::kBool &&
                tree_mask->is_contiguous() &&
                tree_mask->sizes() == (at::IntArrayRef{B, Tq, Tq}),
                "tree_mask must be contiguous bool (B, Tq, Tq)");
    TORCH_CHECK(window == 0, "tree_mask + sliding window: use the torch path");
    CHECK_DEV((*tree_mask));
    // kBool tensors dispatch data_ptr<bool>, not <unsigned char> (Byte)
    tm_ptr = reinterpret_cast<const unsigned char*>(
        tree_mask->data_ptr<bool>());
  }
  auto out = torch::empty_like(q);
  dim3 grid((Tq + 127) / 128, B * Hq);
  auto launch = [&](auto d) {
    attn_prefill_kernel<decltype(d)::value><<<grid, 512, 0, cur_stream()>>>(
        bf_ptr(q), bf_ptr(k_pages), bf_ptr(v_pages), page_table.data_ptr<int>(),
        q_start.data_ptr<int>(), alibi_ptr, tm_ptr, bf_ptr_mut(out), B, Hq, G,
        Tq, P, maxp, (int)window, (float)scale,
        (long)Hq * Tq * D, (long)D, (long)Tq * D,
        (long)Hq * Tq * D, (long)D, (long)Tq * D);
  };
  if (D == 128) launch(std::integral_constant<int, 128>{});
  else if (D == 64) launch(std::integral_constant<int, 64>{});
  else if (D == 256) launch(std::integral_constant<int, 256>{});
  else if (D == 32) launch(std::integral_constant<int, 32>{});
  else TORCH_CHECK(false, "unsupported head_dim ", D);
  return out;
}

// Fused-QKV prefill: q read from qkv (B, Tq, (Hq+2Hkv)*D) in place,
// out (B, Tq, Hq*D) ready for the O-projection GEMM.
torch::Tensor attn_prefill_qkv(torch::Tensor qkv, long Hq_,
                               torch::Tensor k_pages, torch::Tensor v_pages,
                               torch::Tensor page_table, torch::Tensor q_start,
                               double scale, long window) {
  CHECK_DEV(qkv); CHECK_BF16(qkv); CHECK_CONTIG(qkv);
  CHECK_DEV_ALL3(k_pages, page_table, q_start);
  const int Hkv = k_pages.size(1), P = k_pages.size(2), D = k_pages.size(3);
  const int maxp = page_table.size(1);
  const int Hq = (int)Hq_;
  const int B = qkv.size(0), Tq = qkv.size(1);
  const int X = Hq + 2 * Hkv;
  TORCH_CHECK(qkv.dim() == 3 && qkv.size(2) == (long)X * D, "bad fused qkv shape");
  const int G = Hq / Hkv;
  auto out = torch::empty({B, Tq, (long)Hq * D}, qkv.options());
  dim3 grid((Tq + 127) / 128, B * Hq);
  auto launch = [&](auto d) {
    attn_prefill_kernel<decltype(d)::value><<<grid, 512, 0, cur_stream()>>>(
        bf_ptr(qkv), bf_ptr(k_pages), bf_ptr(v_pages), page_table.data_ptr<int>(),
        q_start.data_ptr<int>(), nullptr, nullptr, bf_ptr_mut(out), B, Hq, G,
        Tq, P, maxp, (int)window, (float)scale,
        (long)Tq * X * D, (long)X * D, (long)D,
        (long)Tq * Hq * D, (long)Hq * D, (long)D);
  };
  if (D == 128) launch(std::integral_constant<int, 128>{});
  else if (D == 64) launch(std::integral_constant<int, 64>{});
  else if (D == 256) launch(std::integral_constant<int, 256>{});
  else if (D == 32) launch(std::integral_constant<int, 32>{});
  else TORCH_CHECK(false, "unsupported head_dim ", D);
  return out;
}

// Fused RoPE + paged-KV write on the raw QKV GEMM output.
void rope_kv_write_(torch::Tensor qkv, long Hq_, long Hkv_, torch::Tensor cos_t,
                    torch::Tensor sin_t, c10::optional<torch::Tensor> pos,
                    torch::Tensor k_pages, torch::Tensor v_pages,
                    torch::Tensor page_table, torch::Tensor start_pos) {
  CHECK_DEV(qkv); CHECK_BF16(qkv); CHECK_CONTIG(qkv);
  CHECK_DEV_ALL3(k_pages, page_table, start_pos);
  CHECK_DEV_ALL2(cos_t, sin_t);
  if (pos.has_value()) CHECK_DEV((*pos));
  const int Hq = (int)Hq_, Hkv = (int)Hkv_;
  const int P = k_pages.size(2), D = k_pages.size(3);
  const int maxp = page_table.size(1);
  const int B = qkv.size(0), T = qkv.size(1);
  TORCH_CHECK(qkv.size(2) == (long)(Hq + 2 * Hkv) * D, "bad fused qkv shape");
  const int* pos_ptr = nullptr;
  if (pos.has_value()) {
    TORCH_CHECK(pos->scalar_type() == at::kInt);
    pos_ptr = pos->data_ptr<int>();
  }
  dim3 grid(B * T, Hq + 2 * Hkv);
  rope_kv_write_kernel<<<grid, 64, 0, cur_stream()>>>(
      bf_ptr_mut(qkv), cos_t.data_ptr<float>(), sin_t.data_ptr<float>(), pos_ptr,
      bf_ptr_mut(k_pages), bf_ptr_mut(v_pages), page_table.data_ptr<int>(),
      start_pos.data_ptr<int>(), B, Hq, Hkv, T, D, P, maxp,
      (int)cos_t.size(0));
}


// C(S,N) = A[row(s)] @ W[expert(s)]^T — grouped decode MoE (moe_gemm.hip).
// off: (E+1,) int32 exclusive prefix sums of per-expert slot counts;
// rowmap: optional (S,) int32 slot->A-row gather; scale: optional (S,) f32
// per-slot epilogue scale (routing weights). S and mchunks are host-side
// upper-bound shapes so no device sync is needed.
torch::Tensor moe_gemm(torch::Tensor A, torch::Tensor W, torch::Tensor off,
                       c10::optional<torch::Tensor> rowmap,
                       c10::optional<torch::Tensor> scale,
                       long S, long mchunks) {
  CHECK_DEV(A); CHECK_BF16(A); CHECK_CONTIG(A);
  CHECK_DEV(W); CHECK_BF16(W); CHECK_CONTIG(W);
  CHECK_DEV(off);
  TORCH_CHECK(W.dim() == 3, "W must be (E, N, K)");
  const int E = W.size(0), N = W.size(1), K = W.size(2);
  TORCH_CHECK(A.size(-1) == K, "A/W K mismatch");
  TORCH_CHECK(K % 32 == 0 && N % 64 == 0, "K%32, N%64 required");
  TORCH_CHECK(off.scalar_type() == at::kInt && off.numel() == E + 1);
  const int* rmp = nullptr;
  if (rowmap.has_value()) {
    TORCH_CHECK(rowmap->scalar_type() == at::kInt &&
                rowmap->is_contiguous() && rowmap->numel() == S);
    CHECK_DEV((*rowmap));
    rmp = rowmap->data_ptr<int>();
  }
  const float* sp = nullptr;
  if (scale.has_value()) {
    TORCH_CHECK(scale->scalar_type() == at::kFloat &&
                scale->is_contiguous() && scale->numel() == S);
    CHECK_DEV((*scale));
    sp = scale->data_ptr<float>();
  }
  auto C = torch::empty({S, (long)N}, A.options());
  TORCH_CHECK(mchunks >= 1);
  dim3 grid(N / 64, (unsigned)(E * mchunks));
  if (K % 256 == 0)
    moe_gemm_v2_kernel<2><<<grid, 256, 0, cur_stream()>>>(
        bf_ptr(A), bf_ptr(W), off.data_ptr<int>(), rmp, sp, bf_ptr_mut(C),
        N, K, (int)mchunks);
  else
    moe_gemm_kernel<2><<<grid, 256, 0, cur_stream()>>>(
        bf_ptr(A), bf_ptr(W), off.data_ptr<int>(), rmp, sp, bf_ptr_mut(C),
        N, K, (int)mchunks);
  return C;
}


// C(M,N) = A @ dequant4(Wq)^T [+R] [+bias], M <= 32 (w4_gemm.hip). Wq/scale/
// zero in quant4_pack layout (2 codes/byte along K, f16 scale+zero per group
// of 64). ~3.5x less weight-stream bytes than the bf16 kernel.
torch::Tensor gemm_w4(torch::Tensor A, torch::Tensor Wq, torch::Tensor scale,
                      torch::Tensor zero,
                      c10::optional<torch::Tensor> residual,
                      c10::optional<torch::Tensor> bias, long N_, long ksplit_req) {
  CHECK_DEV(A); CHECK_CONTIG(A);
  TORCH_CHECK(A.scalar_type() == at::kHalf,
              "gemm_w4 takes fp16 activations (wrapper converts)");
  CHECK_DEV_ALL3(Wq, scale, zero);
  CHECK_CONTIG(Wq);
  TORCH_CHECK(Wq.scalar_type() == at::kByte && scale.scalar_type() == at::kHalf
              && zero.scalar_type() == at::kHalf);
  const int K = A.size(-1);
  const int M = A.numel() / K;
  const int N = (int)N_;
  TORCH_CHECK(M <= 32, "gemm_w4 is for M <= 32, got ", M);
  TORCH_CHECK(Wq.numel() == (long)N * K / 2, "Wq shape mismatch");
  TORCH_CHECK(scale.numel() == (long)N * K / 64, "scale shape mismatch");
  TORCH_CHECK(K % 128 == 0 && N % 64 == 0, "K%128, N%64 required");
  auto sizes = A.sizes().vec();
  sizes.back() = N;
  auto C = torch::empty(sizes, A.options().dtype(at::kBFloat16));
  const unsigned short* rp = nullptr;
  if (residual.has_value()) {
    TORCH_CHECK(residual->is_contiguous() && residual->numel() == (long)M * N);
    rp = bf_ptr(*residual);
  }
  const unsigned short* bp = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->is_contiguous() && bias->numel() == N);
    bp = bf_ptr(*bias);
  }
  // M==1 (the speculative-draft GEMV shape): contiguous 1 KB code loads +
  // v_dot2 — measured 2x the bf16 kernel (w4_probe.hip). The MROWS<=4
  // extension was measured and REJECTED for dispatch: amortizing the
  // dequant over rows loses to the split-K MFMA form (M4 down-proj 131 us
  // vs ~13 — the per-lane 4 B A loads scale with M and dominate); the
  // kernel keeps MROWS for completeness (parity-tested).
  if (M == 1 && K % 2048 == 0 && K / 2048 <= 7) {
    const int rpw = 4;
    dim3 gv((N / rpw + 3) / 4);
    auto lv2 = [&](auto nl, auto mr) {
      gemm_w4_gemv_kernel<decltype(nl)::value, decltype(mr)::value>
          <<<gv, 256, 0, cur_stream()>>>(
              reinterpret_cast<const _Float16*>(A.data_ptr()),
              Wq.data_ptr<unsigned char>(),
              reinterpret_cast<const __half*>(scale.data_ptr()),
              reinterpret_cast<const __half*>(zero.data_ptr()),
              rp, bp, bf_ptr_mut(C), N, K, rpw);
    };
    auto lv = [&](auto nl) {
      switch (M) {
        case 1: lv2(nl, std::integral_constant<int, 1>{}); break;
        case 2: lv2(nl, std::integral_constant<int, 2>{}); break;
        case 3: lv2(nl, std::integral_constant<int, 3>{}); break;
        default: lv2(nl, std::integral_constant<int, 4>{}); break;
      }
    };
    switch (K / 2048) {
      case 1: lv(std::integral_constant<int, 1>{}); break;
      case 2: lv(std::integral_constant<int, 2>{}); break;
      case 3: lv(std::integral_constant<int, 3>{}); break;
      case 4: lv(std::integral_constant<int, 4>{}); break;
      case 5: lv(std::integral_constant<int, 5>{}); break;
      case 6: lv(std::integral_constant<int, 6>{}); break;
      default: lv(std::integral_constant<int, 7>{}); break;
    }
    return C;
  }
  int ksplit = (int)ksplit_req;
  if (ksplit <= 0) {
    // the w4 stream is LATENCY-bound at the bf16 kernel's grid size (1.2
    // TB/s at 448 blocks); split K until the grid reaches ~1800 blocks
    // (w4_probe.hip: ksplit=4 on N=28672 -> 2.25 TB/s, plateau beyond)
    ksplit = (int)std::max<long>(1, std::min<long>(K / 512,
                                                   1792 / std::max(1, (N + 63) / 64)));
  }
  // groups of 64 must not straddle splits; slices of 128 keep the k-loop
  // tail-free
  const int nch = K / 128;
  const int per = (nch + ksplit - 1) / ksplit;
  ksplit = (nch + per - 1) / per;
  const int kchunk = per * 128;
  torch::Tensor part;
  float* pp = nullptr;
  if (ksplit > 1) {
    part = torch::empty({(long)ksplit, (long)M, (long)N},
                        torch::TensorOptions().device(A.device()).dtype(at::kFloat));
    pp = part.data_ptr<float>();
  }
  dim3 grid((N + 63) / 64, ksplit);
  auto launch = [&](auto mt) {
    gemm_skinny_w4_kernel<decltype(mt)::value><<<grid, 256, 0, cur_stream()>>>(
        reinterpret_cast<const _Float16*>(A.data_ptr()),
        Wq.data_ptr<unsigned char>(),
        reinterpret_cast<const __half*>(scale.data_ptr()),
        reinterpret_cast<const __half*>(zero.data_ptr()),
        ksplit == 1 ? rp : nullptr, ksplit == 1 ? bp : nullptr,
        bf_ptr_mut(C), pp, M, N, K, kchunk, ksplit);
  };
  if (M <= 16) launch(std::integral_constant<int, 1>{});
  else launch(std::integral_constant<int, 2>{});
  if (ksplit > 1) {
    const long total = (long)M * N;
    const int cgrid = (int)std::min<long>((total + 255) / 256, 2048);
    gemm_skinny_combine_kernel<<<cgrid, 256, 0, cur_stream()>>>(
        pp, rp, bp, bf_ptr_mut(C), M, N, ksplit);
  }
  return C;
}

// ---------------------------------------------------------------------------
// skinny-M GEMM (decode projections)
// ---------------------------------------------------------------------------

// C(M,N) = A(M,K) @ W(N,K)^T [+ residual (M,N)] [+ bias (N,)], M <= 32.
torch::Tensor gemm_skinny(torch::Tensor A, torch::Tensor W,
                          c10::optional<torch::Tensor> residual,
                          c10::optional<torch::Tensor> bias, long ksplit_req,
                          c10::optional<torch::Tensor> norm_w, double eps,
                          c10::optional<torch::Tensor> ss_in,
                          c10::optional<torch::Tensor> ss_out,
                          long norm_mode_req) {
  CHECK_DEV(A); CHECK_BF16(A); CHECK_CONTIG(A);
  CHECK_DEV(W); CHECK_BF16(W); CHECK_CONTIG(W);
  const int K = A.size(-1);
  const int M = A.numel() / K;
  const int N = W.size(0);
  TORCH_CHECK(M <= 32, "gemm_skinny is for M <= 32, got ", M);
  TORCH_CHECK(W.size(1) == K, "A/W K mismatch");
  TORCH_CHECK(K % 32 == 0 && N % 64 == 0, "K%32, N%64 required");
  // norm_mode: 0 none; 1 scale A by invr*norm_w per stage; 2 norm weight
  // pre-folded into W — only the invr accumulator scale remains (cheap).
  int norm_mode = (int)norm_mode_req;
  const unsigned short* nwp = nullptr;
  if (norm_w.has_value()) {
    TORCH_CHECK(norm_w->is_contiguous() && norm_w->numel() == K &&
                norm_w->scalar_type() == at::kBFloat16,
                "norm_w must be contiguous bf16 of length K");
    nwp = bf_ptr(*norm_w);
    if (norm_mode == 0) norm_mode = 1;
  }
  TORCH_CHECK(norm_mode != 1 || nwp, "norm_mode 1 needs norm_w");
  if (norm_mode)
    TORCH_CHECK(K % 256 == 0, "fused rmsnorm needs the v2 (K%256) kernel");
  // ss_in: (nstripes, 32) f32 row sum-of-squares left by the producing
  // GEMM's epilogue — lets the fused rmsnorm skip re-streaming A.
  const float* ssinp = nullptr;
  int nstripes = 0;
  if (ss_in.has_value()) {
    TORCH_CHECK(norm_mode != 0, "ss_in only meaningful with a norm mode");
    TORCH_CHECK(ss_in->is_contiguous() && ss_in->scalar_type() == at::kFloat &&
                ss_in->numel() % 32 == 0, "ss_in must be f32 (nstripes,32)");
    ssinp = ss_in->data_ptr<float>();
    nstripes = (int)(ss_in->numel() / 32);
  }
  float* ssoutp = nullptr;
  if (ss_out.has_value()) {
    TORCH_CHECK(ss_out->is_contiguous() && ss_out->scalar_type() == at::kFloat &&
                ss_out->numel() == (long)(N / 64) * 32,
                "ss_out must be f32 of numel (N/64)*32");
    ssoutp = ss_out->data_ptr<float>();
  }
  auto sizes = A.sizes().vec();
  sizes.back() = N;
  auto C = torch::empty(sizes, A.options());
  const unsigned short* rp = nullptr;
  if (residual.has_value()) {
    TORCH_CHECK(residual->is_contiguous() && residual->numel() == (long)M * N);
    rp = bf_ptr(*residual);
  }
  const unsigned short* bp = nullptr;
  if (bias.has_value()) {
    TORCH_CHECK(bias->is_contiguous() && bias->numel() == N);
    bp = bf_ptr(*bias);
  }
  int ksplit = (int)ksplit_req;
  if (ksplit <= 0) {
    // Measured optima (benchmarks/bench_kernels.py gemm): big-N shapes are
    // BW-bound — once N/64 covers the 256 CUs extra splits only add combine
    // traffic; small-N shapes are per-block-latency-bound and want
    // (N/64)*ksplit ~ 512-1024 blocks, capped at 8.
    // BBAMD_GEMM_KSPLIT_BIG: experiment knob for the in-context latency
    // parking the PMC shows on the big MLP GEMMs (prof_ctx r2k).
    static const int ks_big = [] {
      const char* e = std::getenv("BBAMD_GEMM_KSPLIT_BIG");
      return e ? std::atoi(e) : 1;
    }();
    if (N / 64 >= 256) ksplit = ks_big;
    else ksplit = (int)std::min<long>(K / 512,
                                      std::max<long>(1, std::min<long>(8, 1024 / (N / 64))));
    ksplit = std::max(1, ksplit);
  }
  const bool v2 = (K % 256) == 0;
  int kchunk = K;
  if (v2) {
    // v2 stages 256-element chunks: round the K slice to a stage multiple and
    // drop empty trailing splits.
    const int nch = K / 256;
    const int per = (nch + ksplit - 1) / ksplit;
    ksplit = (nch + per - 1) / per;
    kchunk = per * 256;
  }
  torch::Tensor part;
  float* pp = nullptr;
  if (ksplit > 1) {
    part = torch::empty({(long)ksplit, (long)M, (long)N},
                        torch::TensorOptions().device(A.device()).dtype(at::kFloat));
    pp = part.data_ptr<float>();
  }
  dim3 grid(N / 64, ksplit);
  auto launch = [&](auto mt) {
    if (v2)
      gemm_skinny_v2_kernel<decltype(mt)::value><<<grid, 256, 0, cur_stream()>>>(
          bf_ptr(A), bf_ptr(W), ksplit == 1 ? rp : nullptr,
          ksplit == 1 ? bp : nullptr, bf_ptr_mut(C), pp, M, N, K, kchunk, ksplit,
          nwp, (float)eps, norm_mode, ssinp, nstripes,
          ksplit == 1 ? ssoutp : nullptr);
    else {
      TORCH_CHECK(!norm_mode && (!ssoutp || ksplit > 1),
                  "fused norm / ss_out need the v2 (K%256) kernel");
      gemm_skinny_kernel<decltype(mt)::value><<<grid, 256, 0, cur_stream()>>>(
          bf_ptr(A), bf_ptr(W), ksplit == 1 ? rp : nullptr,
          ksplit == 1 ? bp : nullptr, bf_ptr_mut(C), pp, M, N, K, ksplit);
    }
  };
  if (M <= 16) launch(std::integral_constant<int, 1>{});
  else launch(std::integral_constant<int, 2>{});
  if (ksplit > 1) {
    // the vectorized (float4-per-lane) combine serves both the ss-emitting
    // and plain cases; the flat grid-stride combine remains the fallback
    // for N % 64 != 0 and for narrow N where (N/64, M/16) leaves the grid
    // too small (gemma-class H=2304 -> 36 WGs)
    if (ssoutp || (N % 64 == 0 && N / 64 >= 64)) {
      dim3 cg(N / 64, (M + 15) / 16);
      gemm_skinny_combine_ss_kernel<<<cg, 256, 0, cur_stream()>>>(
          pp, rp, bp, bf_ptr_mut(C), ssoutp, M, N, ksplit);
    } else {
      const long total = (long)M * N;
      const int cgrid = (int)std::min<long>((total + 255) / 256, 2048);
      gemm_skinny_combine_kernel<<<cgrid, 256, 0, cur_stream()>>>(
          pp, rp, bp, bf_ptr_mut(C), M, N, ksplit);
    }
  }
  return C;
}

// ---------------------------------------------------------------------------
// quant4 / selftest
// ---------------------------------------------------------------------------

std::vector<torch::Tensor> quant4_pack(torch::Tensor x) {
  CHECK_DEV(x); CHECK_BF16(x); CHECK_CONTIG(x);
  const long ncols = x.size(-1);
  TORCH_CHECK(ncols % 64 == 0, "quant4 needs cols % 64 == 0");
  const long nrows = x.numel() / ncols;
  auto packed = torch::empty({nrows, ncols / 2},
                             torch::TensorOptions().device(x.device()).dtype(at::kByte));
  auto hopt = torch::TensorOptions().device(x.device()).dtype(at::kHalf);
  auto scale = torch::empty({nrows, ncols / 64}, hopt);
  auto zero = torch::empty({nrows, ncols / 64}, hopt);
  const long ngroups = nrows * (ncols / 64);
  const int wpb = 4;  // waves per block
  quant4_pack_kernel<64><<<(ngroups + wpb - 1) / wpb, wpb * 64, 0, cur_stream()>>>(
      bf_ptr(x), packed.data_ptr<unsigned char>(),
      reinterpret_cast<__half*>(scale.data_ptr()),
      reinterpret_cast<__half*>(zero.data_ptr()), nrows, ncols);
  return {packed, scale, zero};
}

torch::Tensor quant4_unpack(torch::Tensor packed, torch::Tensor scale,
                            torch::Tensor zero) {
  CHECK_DEV(packed);
  const long nrows = packed.size(0), ncols = packed.size(1) * 2;
  auto out = torch::empty({nrows, ncols},
                          torch::TensorOptions().device(packed.device()).dtype(at::kBFloat16));
  const long total = nrows * ncols;
  const int grid = (int)std::min<long>((total + 255) / 256, 4096);
  quant4_unpack_kernel<64><<<grid, 256, 0, cur_stream()>>>(
      packed.data_ptr<unsigned char>(), reinterpret_cast<__half*>(scale.data_ptr()),
      reinterpret_cast<__half*>(zero.data_ptr()), bf_ptr_mut(out), nrows, ncols);
  return out;
}

torch::Tensor kv_stream_probe(torch::Tensor k_pages, torch::Tensor v_pages,
                              torch::Tensor page_table, torch::Tensor ctx_lens,
                              long n_split, bool nt) {
  const int B = page_table.size(0), Hkv = k_pages.size(1);
  const int P = k_pages.size(2), D = k_pages.size(3), maxp = page_table.size(1);
  auto out = torch::empty({(long)B * Hkv * n_split * 4},
                          torch::TensorOptions().device(k_pages.device()).dtype(at::kFloat));
  dim3 grid(B * Hkv, n_split);
  TORCH_CHECK(D == 128);
  if (nt)
    kv_stream_probe_kernel<128, true><<<grid, 256, 0, cur_stream()>>>(
        bf_ptr(k_pages), bf_ptr(v_pages), page_table.data_ptr<int>(),
        ctx_lens.data_ptr<int>(), out.data_ptr<float>(), B, Hkv, P, maxp,
        (int)n_split);
  else
    kv_stream_probe_kernel<128, false><<<grid, 256, 0, cur_stream()>>>(
        bf_ptr(k_pages), bf_ptr(v_pages), page_table.data_ptr<int>(),
        ctx_lens.data_ptr<int>(), out.data_ptr<float>(), B, Hkv, P, maxp,
        (int)n_split);
  return out;
}

torch::Tensor mfma_selftest(torch::Tensor A, torch::Tensor B) {
  CHECK_DEV(A); CHECK_BF16(A); CHECK_CONTIG(A); CHECK_CONTIG(B);
  TORCH_CHECK(A.size(0) == 16 && A.size(1) == 32 && B.size(0) == 32 && B.size(1) == 16);
  auto C = torch::empty({16, 16}, torch::TensorOptions().device(A.device()).dtype(at::kFloat));
  mfma_selftest_kernel<<<1, 64, 0, cur_stream()>>>(bf_ptr(A), bf_ptr(B),
                                                   C.data_ptr<float>());
  return C;
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rms_norm", &rms_norm);
  m.def("rms_norm_residual", &rms_norm_residual);
  m.def("layer_norm", &layer_norm);
  m.def("rope_apply_", &rope_apply_);
  m.def("swiglu", &swiglu);
  m.def("gelu_tanh", &gelu_tanh);
  m.def("add_bf16", &add_bf16);
  m.def("kv_write", &kv_write);
  m.def("kv_gather", &kv_gather);
  m.def("wire_deflate", &wire_deflate);
  m.def("wire_inflate", &wire_inflate);
  m.def("wire_deflate_mt", &wire_deflate_mt);
  m.def("wire_inflate_mt", &wire_inflate_mt);
  m.def("attn_decode", &attn_decode);
  m.def("attn_decode_qkv", &attn_decode_qkv);
  m.def("attn_prefill_qkv", &attn_prefill_qkv);
  m.def("rope_kv_write_", &rope_kv_write_);
  m.def("attn_prefill", &attn_prefill);
  m.def("gemm_skinny", &gemm_skinny, py::arg("A"), py::arg("W"),
        py::arg("residual") = c10::nullopt, py::arg("bias") = c10::nullopt,
        py::arg("ksplit") = 0, py::arg("norm_w") = c10::nullopt,
        py::arg("eps") = 0.0, py::arg("ss_in") = c10::nullopt,
        py::arg("ss_out") = c10::nullopt, py::arg("norm_mode") = 0);
  m.def("moe_gemm", &moe_gemm);
  m.def("gemm_w4", &gemm_w4);
  m.def("quant4_pack", &quant4_pack);
  m.def("quant4_unpack", &quant4_unpack);
  m.def("mfma_selftest", &mfma_selftest);
  m.def("kv_stream_probe", &kv_stream_probe);
}
