// Python bindings for the MI355X (gfx950) HIP kernel library.
//
// Every op here is a hand-written CDNA4 kernel (see the .hip files); the
// Python side (inference_gateway_amd/ops/__init__.py) requires this
// extension whenever a GPU is present — there is no silent eager fallback
// on the GPU path.

#include <torch/extension.h>

namespace igw {
void rmsnorm(torch::Tensor out, torch::Tensor in, torch::Tensor weight,
             double eps);
void fused_add_rmsnorm(torch::Tensor x, torch::Tensor residual,
                       torch::Tensor weight, double eps);
void rotary_embedding(torch::Tensor positions, torch::Tensor q,
                      torch::Tensor k, int64_t head_dim,
                      torch::Tensor cos_sin_cache);
void append_kv_cache(torch::Tensor k, torch::Tensor v, torch::Tensor kcache,
                     torch::Tensor vcache, torch::Tensor slot_mapping);
void paged_attn_decode(torch::Tensor out, torch::Tensor q,
                       torch::Tensor kcache, torch::Tensor vcache,
                       torch::Tensor block_table, torch::Tensor ctx_lens,
                       double scale);
void flash_prefill(torch::Tensor out, torch::Tensor q, torch::Tensor k,
                   torch::Tensor v, torch::Tensor cu_seqlens,
                   torch::Tensor tile_seq, torch::Tensor tile_off,
                   double scale);
void flash_prefill_paged(torch::Tensor out, torch::Tensor q,
                         torch::Tensor kcache, torch::Tensor vcache,
                         torch::Tensor block_table, torch::Tensor prefix_lens,
                         torch::Tensor cu_seqlens, torch::Tensor tile_seq,
                         torch::Tensor tile_off, double scale);
void sample(torch::Tensor out, torch::Tensor logits, torch::Tensor temperature,
            torch::Tensor top_k, torch::Tensor top_p, torch::Tensor uniforms);
void silu_mul(torch::Tensor out, torch::Tensor gate_up);
void moe_route(torch::Tensor weights, torch::Tensor dst,
               torch::Tensor topk_idx, torch::Tensor counts,
               torch::Tensor logits, int64_t C);
void moe_grouped_gemm(torch::Tensor out, torch::Tensor a, torch::Tensor w,
                      torch::Tensor counts, int64_t capacity);
void silu_mul_grouped(torch::Tensor out, torch::Tensor gate_up,
                      torch::Tensor counts, int64_t capacity);
void fused_gemm_tn(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                   bool swiglu, bool nt_weights);
void fused_gemm_tn2(torch::Tensor out, torch::Tensor x, torch::Tensor w,
                    bool swiglu, bool swz);
void glds_probe(torch::Tensor out, torch::Tensor inp, int64_t swizzle);
int64_t ar_alloc(int64_t bytes, bool finegrained);
void ar_free(int64_t ptr);
py::bytes ar_get_handle(int64_t ptr);
int64_t ar_open_handle(py::bytes handle);
void ar_close_handle(int64_t ptr);
int64_t ar_sig_bytes();
int64_t ar_check_error(int64_t sig_ptr);
py::bytes ar_dump(int64_t ptr, int64_t nbytes);
void oneshot_allreduce(torch::Tensor inout, int64_t rank, int64_t world,
                       std::vector<int64_t> data_ptrs,
                       std::vector<int64_t> sig_ptrs,
                       int64_t buf_stride_elems);
}  // namespace igw

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm", &igw::rmsnorm, "RMSNorm (bf16, gfx950)");
  m.def("fused_add_rmsnorm", &igw::fused_add_rmsnorm,
        "residual += x; x = rmsnorm(residual) (bf16, gfx950)");
  m.def("rotary_embedding", &igw::rotary_embedding,
        "NeoX-style RoPE in-place on q/k (bf16, gfx950)");
  m.def("append_kv_cache", &igw::append_kv_cache,
        "scatter step K/V into the paged cache (bf16, gfx950)");
  m.def("paged_attn_decode", &igw::paged_attn_decode,
        "GQA paged-attention decode (bf16, gfx950)");
  m.def("flash_prefill", &igw::flash_prefill,
        "varlen causal MFMA flash-attention prefill (bf16, gfx950)");
  m.def("flash_prefill_paged", &igw::flash_prefill_paged,
        "paged-extend prefill: chunk queries over cached prefix+chunk K/V "
        "via block table (bf16, gfx950)");
  m.def("sample", &igw::sample,
        "fused temperature/top-k/top-p sampling (fp32 logits, gfx950)");
  m.def("silu_mul", &igw::silu_mul, "fused SwiGLU activation (bf16, gfx950)");
  m.def("moe_route", &igw::moe_route,
        "fused MoE top-k routing + softmax + padded-dispatch ranks (gfx950)");
  m.def("moe_grouped_gemm", &igw::moe_grouped_gemm,
        "grouped per-expert MFMA GEMM over the padded dispatch buffer");
  m.def("silu_mul_grouped", &igw::silu_mul_grouped,
        "SwiGLU over only the real (non-padded) MoE dispatch rows");
  m.def("glds_probe", &igw::glds_probe, "glds round-trip probe (debug)");
  m.def("fused_gemm_tn2", &igw::fused_gemm_tn2,
        "TN bf16 MFMA GEMM v2: block-shared LDS weight tiles "
        "(glds + XOR swizzle), optional fused SwiGLU");
  m.def("fused_gemm_tn", &igw::fused_gemm_tn,
        "TN bf16 MFMA GEMM with optional fused SwiGLU epilogue "
        "(decode weight-streaming shapes, gfx950)");
  m.def("ar_alloc", &igw::ar_alloc,
        "raw device allocation for one-shot all-reduce buffers "
        "(finegrained=true for signal pages)");
  m.def("ar_free", &igw::ar_free, "free an ar_alloc allocation");
  m.def("ar_get_handle", &igw::ar_get_handle,
        "hipIpcMemHandle for a raw allocation (64 bytes)");
  m.def("ar_open_handle", &igw::ar_open_handle,
        "open a peer rank's IPC handle -> device pointer");
  m.def("ar_close_handle", &igw::ar_close_handle,
        "close an opened peer IPC handle");
  m.def("ar_sig_bytes", &igw::ar_sig_bytes,
        "size of the one-shot signal page");
  m.def("ar_check_error", &igw::ar_check_error,
        "read the one-shot spin-timeout error word");
  m.def("ar_dump", &igw::ar_dump, "copy raw device bytes to host (debug)");
  m.def("oneshot_allreduce", &igw::oneshot_allreduce,
        "one-shot bf16 sum all-reduce over xGMI peer buffers "
        "(graph-capturable, device-side epochs)");
}
