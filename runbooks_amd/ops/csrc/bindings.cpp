// Python bindings for the runbooks_amd CDNA4 (gfx950) kernel library.

#include <torch/extension.h>

std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps);
std::vector<at::Tensor> rmsnorm_fwd_dec(at::Tensor x, at::Tensor w, double eps);
std::vector<at::Tensor> rmsnorm_res_fwd_dec(at::Tensor x, at::Tensor res, at::Tensor w, double eps);
std::vector<at::Tensor> rmsnorm_res_slab_fwd_dec(at::Tensor x, at::Tensor slabs, at::Tensor w, double eps);
at::Tensor decode_gemm_raw(at::Tensor xs, at::Tensor ws, int64_t M, int64_t N, int64_t K);
at::Tensor lora_delta_(at::Tensor y, at::Tensor t, at::Tensor w, double scale, bool w_transposed);
at::Tensor lora_badd_(at::Tensor y, at::Tensor t, at::Tensor w, double scale);
std::vector<at::Tensor> swiglu_packed_dec(at::Tensor y);
std::vector<at::Tensor> geglu_packed_dec(at::Tensor y);
std::vector<at::Tensor> rmsnorm_bwd(at::Tensor x, at::Tensor w, at::Tensor dy,
                                    at::Tensor inv_rms);
at::Tensor rope_fwd(at::Tensor x, at::Tensor cos, at::Tensor sin,
                    at::Tensor positions, int64_t heads);
at::Tensor rope_bwd(at::Tensor dy, at::Tensor cos, at::Tensor sin,
                    at::Tensor positions, int64_t heads);
void adamw_step(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
                double lr, double beta1, double beta2, double eps, double wd,
                int64_t step);
void adamw_step_multi(at::Tensor table, int64_t nchunks, bool p32, bool g32,
                      double lr, double beta1, double beta2, double eps,
                      double wd, int64_t step);
int64_t adamw_mt_chunk_elems();
at::Tensor sample_tokens(at::Tensor logits, double temperature, int64_t seed);
void kv_append(at::Tensor k, at::Tensor v, at::Tensor k_cache, at::Tensor v_cache,
               at::Tensor slot_mapping);
at::Tensor paged_decode(at::Tensor q, at::Tensor k_cache, at::Tensor v_cache,
                        at::Tensor block_tables, at::Tensor seq_lens,
                        int64_t nsplit, double scale,
                        c10::optional<at::Tensor> seq_starts);
std::vector<at::Tensor> paged_decode_swz(at::Tensor q, at::Tensor k_cache,
                                         at::Tensor v_cache,
                                         at::Tensor block_tables,
                                         at::Tensor seq_lens, int64_t nsplit,
                                         double scale,
                                         c10::optional<at::Tensor> seq_starts);
at::Tensor flash_prefill(at::Tensor q, at::Tensor k, at::Tensor v, double scale);
std::vector<at::Tensor> flash_fwd_train(at::Tensor q, at::Tensor k, at::Tensor v,
                                        double scale);
std::vector<at::Tensor> fa_bwd(at::Tensor dout, at::Tensor q, at::Tensor k,
                               at::Tensor v, at::Tensor out, at::Tensor lse,
                               double scale);
at::Tensor swiglu_fwd(at::Tensor g, at::Tensor u);
std::vector<at::Tensor> swiglu_bwd(at::Tensor dy, at::Tensor g, at::Tensor u);
std::vector<at::Tensor> cross_entropy_fwd(at::Tensor logits, at::Tensor targets,
                                          int64_t ignore_index);
at::Tensor cross_entropy_bwd(at::Tensor logits, at::Tensor targets, at::Tensor lse,
                             double gscale, int64_t ignore_index);
at::Tensor mfma_probe_32x32x16(at::Tensor a, at::Tensor b);
at::Tensor mfma_probe_16x16x32(at::Tensor a, at::Tensor b);
at::Tensor train_gemm_nt(at::Tensor a, at::Tensor b);
at::Tensor skinny_gemm(at::Tensor x, at::Tensor w);
at::Tensor decode_gemm(at::Tensor xs, at::Tensor ws, int64_t M, int64_t N, int64_t K, int64_t force_split);
at::Tensor decode_swizzle_w(at::Tensor w);
at::Tensor decode_swizzle_x(at::Tensor x);
int64_t decode_gemm_split(int64_t N, int64_t K);
bool decode_gemm_supported(int64_t M, int64_t N, int64_t K);
at::Tensor skinny_gemm_fp8(at::Tensor x, at::Tensor w8, at::Tensor scale);
int64_t skinny_gemm_mmax();
at::Tensor qkv_rope_append(at::Tensor y, at::Tensor cos, at::Tensor sin,
                           at::Tensor positions, at::Tensor k_cache,
                           at::Tensor v_cache, at::Tensor slot_mapping,
                           int64_t hq);
at::Tensor swiglu_packed(at::Tensor y);
at::Tensor geglu_packed(at::Tensor y);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "runbooks_amd gfx950 HIP kernels";
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "fused RMSNorm forward (bf16/f32)");
  m.def("rmsnorm_fwd_dec", &rmsnorm_fwd_dec,
        "RMSNorm forward + decode-GEMM operand swizzle in one pass");
  m.def("rmsnorm_res_fwd_dec", &rmsnorm_res_fwd_dec,
        "residual add + RMSNorm + decode operand swizzle in one pass");
  m.def("rmsnorm_res_slab_fwd_dec", &rmsnorm_res_slab_fwd_dec,
        "split-K slab combine + residual + RMSNorm + swizzle in one pass");
  m.def("decode_gemm_raw", &decode_gemm_raw,
        "decode GEMM returning the uncombined fp32 slab at split>1");
  m.def("lora_delta_", &lora_delta_,
        "in-place y += scale * t[T,r<=32] @ W[N,r]^T (LoRA merge)");
  m.def("lora_badd_", &lora_badd_,
        "MFMA in-place y += scale * t[T,16] @ W[N,16]^T (LoRA B merge)");
  m.def("swiglu_packed_dec", &swiglu_packed_dec,
        "packed SwiGLU + decode-GEMM operand swizzle in one pass");
  m.def("geglu_packed_dec", &geglu_packed_dec,
        "packed GeGLU + decode-GEMM operand swizzle in one pass");
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "fused RMSNorm backward");
  m.def("rope_fwd", &rope_fwd, "RoPE rotate-half forward");
  m.def("rope_bwd", &rope_bwd, "RoPE rotate-half backward (inverse rotation)");
  m.def("adamw_step", &adamw_step, "fused AdamW single-tensor step");
  m.def("adamw_step_multi", &adamw_step_multi,
        "fused AdamW multi-tensor step over a cached chunk table");
  m.def("adamw_mt_chunk_elems", &adamw_mt_chunk_elems,
        "elements per multi-tensor chunk");
  m.def("sample_tokens", &sample_tokens, "greedy / Gumbel-max sampling");
  m.def("kv_append", &kv_append, "paged KV-cache append");
  m.def("paged_decode", &paged_decode, "paged GQA/MQA decode attention",
        py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
        py::arg("block_tables"), py::arg("seq_lens"), py::arg("nsplit"),
        py::arg("scale"), py::arg("seq_starts") = py::none());
  m.def("paged_decode_swz", &paged_decode_swz,
        "paged decode also emitting the o_proj decode-GEMM operand",
        py::arg("q"), py::arg("k_cache"), py::arg("v_cache"),
        py::arg("block_tables"), py::arg("seq_lens"), py::arg("nsplit"),
        py::arg("scale"), py::arg("seq_starts") = py::none());
  m.def("flash_prefill", &flash_prefill, "MFMA flash-attention prefill (causal)");
  m.def("flash_fwd_train", &flash_fwd_train,
        "MFMA flash-attention forward + log-sum-exp (training)");
  m.def("fa_bwd", &fa_bwd,
        "MFMA flash-attention backward -> (dq, dk_perq, dv_perq)");
  m.def("swiglu_fwd", &swiglu_fwd, "fused silu(g)*u");
  m.def("swiglu_bwd", &swiglu_bwd, "fused SwiGLU backward");
  m.def("cross_entropy_fwd", &cross_entropy_fwd, "fused CE: per-row loss + lse");
  m.def("cross_entropy_bwd", &cross_entropy_bwd, "fused CE backward -> dlogits");
  m.def("mfma_probe_16x16x32", &mfma_probe_16x16x32,
        "layout probe: one 16x16x32 bf16 MFMA as a plain matmul");
  m.def("mfma_probe_32x32x16", &mfma_probe_32x32x16,
        "layout probe: one 32x32x16 bf16 MFMA as a plain matmul");
  m.def("skinny_gemm", &skinny_gemm,
        "HBM-rate decode GEMM: y[M<=32,N] = x @ W^T (bf16)");
  m.def("skinny_gemm_mmax", &skinny_gemm_mmax);
  m.def("decode_gemm", &decode_gemm,
        "v2 weight-stream decode GEMM over pre-swizzled operands: "
        "y[M<=32,N] = x @ W^T (bf16), 1KiB coalesced nt weight bursts",
        py::arg("xs"), py::arg("ws"), py::arg("M"), py::arg("N"),
        py::arg("K"), py::arg("force_split") = -1);
  m.def("decode_swizzle_w", &decode_swizzle_w,
        "one-time [N,K] -> fragment-lane-major weight layout");
  m.def("decode_swizzle_x", &decode_swizzle_x,
        "per-call x -> B-fragment lane order (zero-padded to M=32)");
  m.def("decode_gemm_split", &decode_gemm_split);
  m.def("decode_gemm_supported", &decode_gemm_supported);
  m.def("mfma_probe_16x16x32", &mfma_probe_16x16x32,
        "layout probe: one 16x16x32 bf16 MFMA as a plain matmul");
  m.def("train_gemm_nt", &train_gemm_nt,
        "EXPERIMENTAL 256^2 8-phase bf16 GEMM (C = A @ B^T)");
  m.def("skinny_gemm_fp8", &skinny_gemm_fp8,
        "fp8-e4m3 weight-only decode GEMM with per-channel scales");
  m.def("qkv_rope_append", &qkv_rope_append,
        "packed qkv -> rope'd q + rope'd k / v appended to the paged cache");
  m.def("swiglu_packed", &swiglu_packed,
        "silu(y[:, :I]) * y[:, I:] from the fused gate/up GEMM output");
  m.def("geglu_packed", &geglu_packed,
        "packed GeGLU (tanh) epilogue [T,2I]->[T,I]");
}
