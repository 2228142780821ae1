// Torch bindings for the wva_amd MI355X (gfx950) kernels.
//
// Built in-tree via torch.utils.cpp_extension (PYTORCH_ROCM_ARCH=gfx950);
// the .so travels with the repo snapshot to GPU boxes.

#include <torch/extension.h>
#include <hip/hip_runtime.h>
#include <c10/hip/HIPStream.h>

extern "C" void launch_rmsnorm(void* out, void* residual_out,
                               const void* input, const void* residual,
                               const void* weight, float eps, int rows,
                               int hidden, hipStream_t stream);
extern "C" void launch_rope(void* q, void* k, const int* positions, int tokens,
                            int num_q_heads, int num_k_heads, int head_dim,
                            float theta, hipStream_t stream);
extern "C" void launch_silu_mul(void* out, const void* gate, const void* up,
                                long n, hipStream_t stream);
extern "C" void launch_silu_mul_fused(void* out, const void* gate_up,
                                      long rows, long inter,
                                      hipStream_t stream);
extern "C" void launch_rope_append_kv_fp8(const void* qkv, void* q_out,
                                          void* k_cache, void* v_cache,
                                          const int* positions, int batch,
                                          int num_q_heads, int num_kv_heads,
                                          int head_dim, int max_seq,
                                          float theta, hipStream_t stream);
extern "C" void launch_gqa_decode_attn_v4_ex(void* out, void* workspace,
                                             const void* q,
                                             const void* k_cache,
                                             const void* v_cache,
                                             const int* context_lens,
                                             int batch, int num_q_heads,
                                             int num_kv_heads, int max_seq,
                                             int num_splits, float scale,
                                             int kv_fp8, hipStream_t stream);
extern "C" void launch_gqa_decode_attn_v5_ex(void* out, void* workspace,
                                             const void* q,
                                             const void* k_cache,
                                             const void* v_cache,
                                             const int* context_lens,
                                             int batch, int num_q_heads,
                                             int num_kv_heads, int max_seq,
                                             int num_splits, float scale,
                                             int kv_fp8, hipStream_t stream);
extern "C" void launch_rope_append_kv(const void* qkv, void* q_out,
                                      void* k_cache, void* v_cache,
                                      const int* positions, int batch,
                                      int num_q_heads, int num_kv_heads,
                                      int head_dim, int max_seq, float theta,
                                      hipStream_t stream);
extern "C" int gqa_decode_attn_num_splits(int batch, int num_kv_heads,
                                          int max_ctx_hint);
extern "C" int skinny_gemm_num_splits(int N, int K, int nt);
extern "C" void launch_prefill_attn_ex(void* out, const void* q,
                                       const void* k_cache,
                                       const void* v_cache, int batch,
                                       int seq, int num_q_heads,
                                       int num_kv_heads, int max_seq,
                                       float scale, int kv_fp8,
                                       hipStream_t stream);
extern "C" int skinny_gemm_tile_n(int M);
extern "C" void launch_skinny_gemm(void* c, void* ws, const void* a,
                                   const void* w, int M, int N, int K,
                                   int num_splits, hipStream_t stream);
extern "C" void launch_skinny_gemm_ex(void* c, void* ws, const void* a,
                                      const void* w, const float* w_scale,
                                      int M, int N, int K, int num_splits,
                                      int w_fp8, hipStream_t stream);
extern "C" void launch_gqa_decode_attn(void* out, void* workspace,
                                       const void* q, const void* k_cache,
                                       const void* v_cache,
                                       const int* context_lens, int batch,
                                       int num_q_heads, int num_kv_heads,
                                       int max_seq, int num_splits,
                                       float scale, hipStream_t stream);
extern "C" void launch_gqa_decode_attn_v5(void* out, void* workspace,
                                          const void* q, const void* k_cache,
                                          const void* v_cache,
                                          const int* context_lens, int batch,
                                          int num_q_heads, int num_kv_heads,
                                          int max_seq, int num_splits,
                                          float scale, hipStream_t stream);
extern "C" void launch_gqa_decode_attn_v4(void* out, void* workspace,
                                          const void* q, const void* k_cache,
                                          const void* v_cache,
                                          const int* context_lens, int batch,
                                          int num_q_heads, int num_kv_heads,
                                          int max_seq, int num_splits,
                                          float scale, hipStream_t stream);

namespace {

hipStream_t current_stream() {
  return c10::hip::getCurrentHIPStream().stream();
}

void check_bf16_contig(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.scalar_type() == torch::kBFloat16, name, " must be bf16");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
}

// KV caches may be bf16 or fp8 e4m3 (torch float8_e4m3fn); returns true
// when the cache is fp8.
bool check_cache_contig(const torch::Tensor& t, const char* name) {
  TORCH_CHECK(t.is_cuda(), name, " must be on GPU");
  TORCH_CHECK(t.is_contiguous(), name, " must be contiguous");
  if (t.scalar_type() == torch::kBFloat16) return false;
  TORCH_CHECK(t.scalar_type() == at::kFloat8_e4m3fn,
              name, " must be bf16 or float8_e4m3fn");
  return true;
}

// out = rmsnorm(input [+ residual]) * weight; when residual is given it is
// updated in place to (input + residual).
torch::Tensor rmsnorm(torch::Tensor input, torch::Tensor weight,
                      c10::optional<torch::Tensor> residual, double eps) {
  check_bf16_contig(input, "input");
  check_bf16_contig(weight, "weight");
  const int hidden = input.size(-1);
  TORCH_CHECK(hidden % 8 == 0, "hidden must be a multiple of 8");
  TORCH_CHECK(weight.numel() == hidden, "weight size mismatch");
  const long rows = input.numel() / hidden;
  auto out = torch::empty_like(input);
  const void* res_ptr = nullptr;
  void* res_out_ptr = nullptr;
  if (residual.has_value()) {
    check_bf16_contig(residual.value(), "residual");
    TORCH_CHECK(residual->sizes() == input.sizes(), "residual shape mismatch");
    res_ptr = residual->data_ptr();
    res_out_ptr = residual->data_ptr();  // in-place fold
  }
  launch_rmsnorm(out.data_ptr(), res_out_ptr, input.data_ptr(), res_ptr,
                 weight.data_ptr(), (float)eps, (int)rows, hidden,
                 current_stream());
  return out;
}

// In-place NeoX-style RoPE on q [T,Hq,D] and k [T,Hk,D] at positions [T].
void rope(torch::Tensor q, torch::Tensor k, torch::Tensor positions,
          double theta) {
  check_bf16_contig(q, "q");
  check_bf16_contig(k, "k");
  TORCH_CHECK(positions.is_cuda() && positions.scalar_type() == torch::kInt32,
              "positions must be int32 on GPU");
  TORCH_CHECK(q.dim() == 3 && k.dim() == 3, "q/k must be [T, H, D]");
  const int tokens = q.size(0);
  const int num_q_heads = q.size(1);
  const int num_k_heads = k.size(1);
  const int head_dim = q.size(2);
  TORCH_CHECK(k.size(0) == tokens && k.size(2) == head_dim, "q/k mismatch");
  TORCH_CHECK(head_dim % 2 == 0, "head_dim must be even");
  launch_rope(q.data_ptr(), k.data_ptr(), positions.data_ptr<int>(), tokens,
              num_q_heads, num_k_heads, head_dim, (float)theta,
              current_stream());
}

torch::Tensor silu_mul(torch::Tensor gate, torch::Tensor up) {
  check_bf16_contig(gate, "gate");
  check_bf16_contig(up, "up");
  TORCH_CHECK(gate.sizes() == up.sizes(), "gate/up shape mismatch");
  TORCH_CHECK(gate.numel() % 2 == 0, "numel must be even");
  auto out = torch::empty_like(gate);
  launch_silu_mul(out.data_ptr(), gate.data_ptr(), up.data_ptr(),
                  (long)gate.numel(), current_stream());
  return out;
}

// Decode fast path: strided qkv row → RoPE'd q_out + cache append.
torch::Tensor rope_append_kv(torch::Tensor qkv, torch::Tensor k_cache,
                             torch::Tensor v_cache, torch::Tensor positions,
                             int64_t num_q_heads, int64_t num_kv_heads,
                             double theta) {
  check_bf16_contig(qkv, "qkv");
  const bool fp8 = check_cache_contig(k_cache, "k_cache");
  TORCH_CHECK(check_cache_contig(v_cache, "v_cache") == fp8,
              "k/v cache dtype mismatch");
  TORCH_CHECK(positions.is_cuda() && positions.scalar_type() == torch::kInt32,
              "positions must be int32 on GPU");
  TORCH_CHECK(qkv.dim() == 2, "qkv must be [B, (Hq+2Hk)*D]");
  TORCH_CHECK(k_cache.dim() == 4, "k_cache must be [B, Hk, S, D]");
  const int batch = qkv.size(0);
  const int head_dim = k_cache.size(3);
  const int max_seq = k_cache.size(2);
  TORCH_CHECK(k_cache.size(1) == num_kv_heads, "Hk mismatch");
  TORCH_CHECK(qkv.size(1) == (num_q_heads + 2 * num_kv_heads) * head_dim,
              "qkv width mismatch");
  auto q_out = torch::empty({batch, num_q_heads, head_dim}, qkv.options());
  auto launch = fp8 ? launch_rope_append_kv_fp8 : launch_rope_append_kv;
  launch(qkv.data_ptr(), q_out.data_ptr(), k_cache.data_ptr(),
         v_cache.data_ptr(), positions.data_ptr<int>(), batch,
         (int)num_q_heads, (int)num_kv_heads, head_dim, max_seq,
         (float)theta, current_stream());
  return q_out;
}

torch::Tensor silu_mul_fused(torch::Tensor gate_up) {
  check_bf16_contig(gate_up, "gate_up");
  TORCH_CHECK(gate_up.dim() == 2 && gate_up.size(1) % 4 == 0,
              "gate_up must be [rows, 2*inter], inter even");
  const long rows = gate_up.size(0);
  const long inter = gate_up.size(1) / 2;
  auto out = torch::empty({rows, inter}, gate_up.options());
  launch_silu_mul_fused(out.data_ptr(), gate_up.data_ptr(), rows, inter,
                        current_stream());
  return out;
}

torch::Tensor gqa_decode_attn_impl(torch::Tensor q, torch::Tensor k_cache,
                                   torch::Tensor v_cache,
                                   torch::Tensor context_lens, double scale,
                                   int use_v4) {
  check_bf16_contig(q, "q");
  const bool kv_fp8 = check_cache_contig(k_cache, "k_cache");
  TORCH_CHECK(check_cache_contig(v_cache, "v_cache") == kv_fp8,
              "k/v cache dtype mismatch");
  TORCH_CHECK(!(kv_fp8 && use_v4 == 0),
              "fp8 KV cache requires the v4/v5 MFMA kernels");
  TORCH_CHECK(context_lens.is_cuda() &&
                  context_lens.scalar_type() == torch::kInt32,
              "context_lens must be int32 on GPU");
  TORCH_CHECK(q.dim() == 3, "q must be [B, Hq, D]");
  TORCH_CHECK(k_cache.dim() == 4,
              "k_cache must be [B, Hk, S, D] (head-major)");
  const int batch = q.size(0);
  const int num_q_heads = q.size(1);
  const int head_dim = q.size(2);
  const int num_kv_heads = k_cache.size(1);
  const int max_seq = k_cache.size(2);
  TORCH_CHECK(head_dim == 128, "head_dim must be 128 (Llama-3 family)");
  TORCH_CHECK(num_q_heads % num_kv_heads == 0, "Hq must divide by Hk");
  TORCH_CHECK(num_q_heads / num_kv_heads <= 8, "GQA group size must be <= 8");
  TORCH_CHECK(v_cache.sizes() == k_cache.sizes(), "k/v cache mismatch");
  auto out = torch::empty_like(q);
  const int num_splits =
      gqa_decode_attn_num_splits(batch, num_kv_heads, max_seq);
  torch::Tensor workspace;
  void* ws_ptr = nullptr;
  if (num_splits > 1) {
    const int G = num_q_heads / num_kv_heads;
    workspace = torch::empty(
        {(long)batch * num_kv_heads * G * num_splits * (2 + head_dim)},
        q.options().dtype(torch::kFloat32));
    ws_ptr = workspace.data_ptr();
  }
  if (use_v4 == 0) {
    launch_gqa_decode_attn(out.data_ptr(), ws_ptr, q.data_ptr(),
                           k_cache.data_ptr(), v_cache.data_ptr(),
                           context_lens.data_ptr<int>(), batch, num_q_heads,
                           num_kv_heads, max_seq, num_splits, (float)scale,
                           current_stream());
  } else {
    auto launch = use_v4 == 2 ? launch_gqa_decode_attn_v5_ex
                              : launch_gqa_decode_attn_v4_ex;
    launch(out.data_ptr(), ws_ptr, q.data_ptr(),
           k_cache.data_ptr(), v_cache.data_ptr(),
           context_lens.data_ptr<int>(), batch, num_q_heads,
           num_kv_heads, max_seq, num_splits, (float)scale,
           kv_fp8 ? 1 : 0, current_stream());
  }
  return out;
}

// Default dispatch, measured per shape class (profiles/attn_v4_ab.txt):
// v5 (MFMA scores + MFMA PV) wins when the PV matrix is well-utilized
// (G = 8) or the grid is unsplit (large batch); v4 (MFMA scores, vector
// PV) wins for G < 8 with split-KV where v5's padded 16-row PV and its
// third per-tile barrier cost more than the VALU sweep it replaces.
torch::Tensor gqa_decode_attn(torch::Tensor q, torch::Tensor k_cache,
                              torch::Tensor v_cache,
                              torch::Tensor context_lens, double scale) {
  const int G = (int)(q.size(1) / k_cache.size(1));
  const int splits = gqa_decode_attn_num_splits(
      (int)q.size(0), (int)k_cache.size(1), (int)k_cache.size(2));
  const int variant = (G >= 8 || splits == 1) ? 2 : 1;
  return gqa_decode_attn_impl(q, k_cache, v_cache, context_lens, scale,
                              variant);
}

// v5 (MFMA scores + MFMA PV) — A/B variant until measured faster.
torch::Tensor gqa_decode_attn_v5(torch::Tensor q, torch::Tensor k_cache,
                                 torch::Tensor v_cache,
                                 torch::Tensor context_lens, double scale) {
  return gqa_decode_attn_impl(q, k_cache, v_cache, context_lens, scale,
                              /*use_v4=*/2);
}

torch::Tensor gqa_decode_attn_v3(torch::Tensor q, torch::Tensor k_cache,
                                 torch::Tensor v_cache,
                                 torch::Tensor context_lens, double scale) {
  return gqa_decode_attn_impl(q, k_cache, v_cache, context_lens, scale,
                              /*use_v4=*/0);
}

// Decode linear: y[M,N] = x[M,K] @ w[N,K]^T on the weight-streaming
// MFMA kernel (csrc/skinny_gemm.hip). Caller guarantees M <= 64 and
// K % 128 == 0; larger M belongs to hipBLASLt.
torch::Tensor skinny_linear(torch::Tensor x, torch::Tensor w) {
  check_bf16_contig(x, "x");
  check_bf16_contig(w, "w");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2, "x [M,K], w [N,K]");
  const int M = x.size(0);
  const int K = x.size(1);
  const int N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "K mismatch");
  TORCH_CHECK(M <= 64, "skinny_linear requires M <= 64");
  TORCH_CHECK(K % 128 == 0, "K must be a multiple of 128");
  auto y = torch::empty({M, N}, x.options());
  const int nt = skinny_gemm_tile_n(M);
  const int sk = skinny_gemm_num_splits(N, K, nt);
  torch::Tensor ws;
  void* ws_ptr = nullptr;
  if (sk > 1) {
    ws = torch::empty({(long)sk * M * N},
                      x.options().dtype(torch::kFloat32));
    ws_ptr = ws.data_ptr();
  }
  launch_skinny_gemm(y.data_ptr(), ws_ptr, x.data_ptr(), w.data_ptr(), M, N,
                     K, sk, current_stream());
  return y;
}

// Weight-only fp8 (W8A16): w is [N, K] e4m3 with per-channel scales [N];
// activations stay bf16, dequantization happens in-fragment and the
// scale folds into the store (csrc/skinny_gemm.hip W_FP8).
torch::Tensor skinny_linear_fp8(torch::Tensor x, torch::Tensor w,
                                torch::Tensor w_scale) {
  check_bf16_contig(x, "x");
  TORCH_CHECK(w.is_cuda() && w.is_contiguous() &&
                  w.scalar_type() == at::kFloat8_e4m3fn,
              "w must be contiguous float8_e4m3fn");
  TORCH_CHECK(w_scale.is_cuda() && w_scale.scalar_type() == torch::kFloat &&
                  w_scale.numel() == w.size(0),
              "w_scale must be float32 [N]");
  TORCH_CHECK(x.dim() == 2 && w.dim() == 2, "x [M,K], w [N,K]");
  const int M = x.size(0);
  const int K = x.size(1);
  const int N = w.size(0);
  TORCH_CHECK(w.size(1) == K, "K mismatch");
  TORCH_CHECK(M <= 64, "skinny_linear requires M <= 64");
  TORCH_CHECK(K % 128 == 0, "K must be a multiple of 128");
  auto y = torch::empty({M, N}, x.options());
  const int nt = skinny_gemm_tile_n(M);
  const int sk = skinny_gemm_num_splits(N, K, nt);
  torch::Tensor ws;
  void* ws_ptr = nullptr;
  if (sk > 1) {
    ws = torch::empty({(long)sk * M * N},
                      x.options().dtype(torch::kFloat32));
    ws_ptr = ws.data_ptr();
  }
  launch_skinny_gemm_ex(y.data_ptr(), ws_ptr, x.data_ptr(), w.data_ptr(),
                        w_scale.data_ptr<float>(), M, N, K, sk, 1,
                        current_stream());
  return y;
}

// Causal flash-attention prefill over the (already-populated) head-major
// KV caches (csrc/prefill_attention.hip). q/out: [B*S, Hq, 128].
torch::Tensor prefill_attn(torch::Tensor q, torch::Tensor k_cache,
                           torch::Tensor v_cache, int64_t batch, int64_t seq,
                           double scale) {
  check_bf16_contig(q, "q");
  const bool kv_fp8 = check_cache_contig(k_cache, "k_cache");
  TORCH_CHECK(check_cache_contig(v_cache, "v_cache") == kv_fp8,
              "k/v cache dtype mismatch");
  TORCH_CHECK(q.dim() == 3 && q.size(2) == 128, "q must be [T, Hq, 128]");
  TORCH_CHECK(q.size(0) == batch * seq, "T must equal B*S");
  TORCH_CHECK(k_cache.dim() == 4, "k_cache must be [B, Hk, S_max, D]");
  const int num_q_heads = q.size(1);
  const int num_kv_heads = k_cache.size(1);
  const int max_seq = k_cache.size(2);
  TORCH_CHECK(seq <= max_seq, "seq exceeds cache size");
  TORCH_CHECK(num_q_heads % num_kv_heads == 0, "Hq must divide by Hk");
  TORCH_CHECK(num_q_heads / num_kv_heads <= 8, "GQA group size must be <= 8");
  auto out = torch::empty_like(q);
  launch_prefill_attn_ex(out.data_ptr(), q.data_ptr(), k_cache.data_ptr(),
                         v_cache.data_ptr(), (int)batch, (int)seq,
                         num_q_heads, num_kv_heads, max_seq, (float)scale,
                         kv_fp8 ? 1 : 0, current_stream());
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &rmsnorm, "Fused (add-)RMSNorm bf16",
        py::arg("input"), py::arg("weight"), py::arg("residual") = py::none(),
        py::arg("eps") = 1e-5);
  m.def("rope", &rope, "Fused in-place NeoX RoPE on q,k",
        py::arg("q"), py::arg("k"), py::arg("positions"),
        py::arg("theta") = 500000.0);
  m.def("silu_mul", &silu_mul, "Fused SwiGLU silu(gate)*up",
        py::arg("gate"), py::arg("up"));
  m.def("silu_mul_fused", &silu_mul_fused,
        "SwiGLU on the fused [rows, 2*inter] gate_up buffer",
        py::arg("gate_up"));
  m.def("rope_append_kv", &rope_append_kv,
        "RoPE on strided qkv row + KV-cache append; returns contiguous q",
        py::arg("qkv"), py::arg("k_cache"), py::arg("v_cache"),
        py::arg("positions"), py::arg("num_q_heads"), py::arg("num_kv_heads"),
        py::arg("theta") = 500000.0);
  m.def("gqa_decode_attn", &gqa_decode_attn,
        "GQA decode attention over contiguous KV cache",
        py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
        py::arg("context_lens"), py::arg("scale"));
  m.def("gqa_decode_attn_v4", &gqa_decode_attn,
        "GQA decode attention, MFMA-scores variant (= default)",
        py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
        py::arg("context_lens"), py::arg("scale"));
  m.def("prefill_attn", &prefill_attn,
        "Causal flash-attention prefill over head-major KV caches",
        py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
        py::arg("batch"), py::arg("seq"), py::arg("scale"));
  m.def("skinny_linear_fp8", &skinny_linear_fp8,
        "Weight-only fp8 decode GEMM: x[M,K]bf16 @ w[N,K]e4m3^T, M <= 64",
        py::arg("x"), py::arg("w"), py::arg("w_scale"));
  m.def("skinny_linear", &skinny_linear,
        "Weight-streaming decode GEMM: x[M,K] @ w[N,K]^T, M <= 64",
        py::arg("x"), py::arg("w"));
  m.def("gqa_decode_attn_v5", &gqa_decode_attn_v5,
        "GQA decode attention, MFMA scores + MFMA PV variant",
        py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
        py::arg("context_lens"), py::arg("scale"));
  m.def("gqa_decode_attn_v3", &gqa_decode_attn_v3,
        "GQA decode attention, pre-MFMA shared-tile variant",
        py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
        py::arg("context_lens"), py::arg("scale"));
}
