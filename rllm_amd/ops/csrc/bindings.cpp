// Python bindings for the rllm_amd CDNA4 kernel library (_C extension).
#include <torch/extension.h>

// elementwise.hip
torch::Tensor add_rmsnorm_(torch::Tensor h, c10::optional<torch::Tensor> delta,
                           torch::Tensor w, double eps);
std::vector<torch::Tensor> rmsnorm_fwd(torch::Tensor x, torch::Tensor w, double eps, bool save_inv_rms);
std::vector<torch::Tensor> rmsnorm_bwd(torch::Tensor dy, torch::Tensor x, torch::Tensor w, torch::Tensor inv_rms);
void rope_inplace(torch::Tensor q, torch::Tensor k, torch::Tensor cos_tab, torch::Tensor sin_tab,
                  torch::Tensor positions, bool backward);
torch::Tensor swiglu_fwd(torch::Tensor gateup);
torch::Tensor swiglu_bwd(torch::Tensor dout, torch::Tensor gateup);

// logprob.hip
void lse_chunk_update(torch::Tensor logits, torch::Tensor m_state, torch::Tensor s_state,
                      c10::optional<torch::Tensor> e_state, torch::Tensor target_logit,
                      torch::Tensor targets, int64_t chunk_start, double inv_temp);
torch::Tensor ce_bwd_chunk(torch::Tensor logits, torch::Tensor lse, torch::Tensor dlp,
                           torch::Tensor targets, int64_t chunk_start, double inv_temp);

// grpo.hip
std::vector<torch::Tensor> grpo_loss_fwd(torch::Tensor logprob, torch::Tensor old_logprob,
                                         c10::optional<torch::Tensor> ref_logprob, torch::Tensor advantages,
                                         c10::optional<torch::Tensor> tis_w,
                                         double eps_lo, double eps_hi, double kl_beta);

// adamw.hip
torch::Tensor grad_sq_sum(torch::Tensor grad, double grad_scale);
void adamw_step(torch::Tensor grad, torch::Tensor master, torch::Tensor m, torch::Tensor v,
                torch::Tensor param, c10::optional<torch::Tensor> gnorm_sq,
                double lr, double beta1, double beta2, double eps, double weight_decay,
                int64_t step, double grad_clip, double grad_scale);

// attention.hip
torch::Tensor flash_prefill(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                            torch::Tensor tile_seq_start, torch::Tensor tile_row0,
                            torch::Tensor tile_seq_len, double scale);
std::vector<torch::Tensor> flash_train_fwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                           torch::Tensor tile_seq_start, torch::Tensor tile_row0,
                                           torch::Tensor tile_seq_len, double scale);
std::vector<torch::Tensor> flash_train_bwd(torch::Tensor q, torch::Tensor k, torch::Tensor v,
                                           torch::Tensor o, torch::Tensor dout, torch::Tensor lse,
                                           torch::Tensor tile_seq_start, torch::Tensor tile_row0,
                                           torch::Tensor tile_seq_len, double scale);
torch::Tensor paged_decode(torch::Tensor q, torch::Tensor k_pages, torch::Tensor v_pages,
                           torch::Tensor block_tables, torch::Tensor seq_lens, double scale,
                           int64_t n_splits);
std::vector<torch::Tensor> gather_cache(torch::Tensor k_pages, torch::Tensor v_pages,
                                        torch::Tensor slot_mapping);
void reshape_and_cache(torch::Tensor k, torch::Tensor v, torch::Tensor k_pages,
                       torch::Tensor v_pages, torch::Tensor slot_mapping);
torch::Tensor qkv_rope_cache(torch::Tensor qkv, c10::optional<torch::Tensor> bias,
                             torch::Tensor k_pages, torch::Tensor v_pages,
                             torch::Tensor cos_tab, torch::Tensor sin_tab,
                             torch::Tensor positions, torch::Tensor slot_mapping,
                             int64_t Hq, int64_t Hk);

// skinny_gemm.hip
torch::Tensor skinny_gemm(torch::Tensor a, torch::Tensor w, c10::optional<torch::Tensor> bias);
torch::Tensor skinny_gemm_v3(torch::Tensor a, torch::Tensor w, c10::optional<torch::Tensor> bias);

// hbl_tuned.hip
std::vector<double> hbl_tune(torch::Tensor x, torch::Tensor w, int64_t iters);
bool hbl_has(int64_t M, int64_t N, int64_t K);
torch::Tensor hbl_mm(torch::Tensor x, torch::Tensor w);
std::vector<double> hbl_fp8_tune(torch::Tensor x8, torch::Tensor w8, torch::Tensor sx, torch::Tensor sw, int64_t iters);
bool hbl_fp8_has(int64_t M, int64_t N, int64_t K);
torch::Tensor hbl_fp8_mm(torch::Tensor x8, torch::Tensor w8, torch::Tensor sx, torch::Tensor sw);

// elementwise.hip (fp8)
void fp8_quant_delayed(torch::Tensor x, torch::Tensor y, torch::Tensor scale, torch::Tensor amax_next);
void fp8_scale_update(torch::Tensor scale, torch::Tensor amax_next);
torch::Tensor add_rmsnorm_fp8_(torch::Tensor h, c10::optional<torch::Tensor> delta,
                               torch::Tensor w, double eps,
                               torch::Tensor scale, torch::Tensor amax_next);
torch::Tensor swiglu_fp8(torch::Tensor gateup, torch::Tensor scale, torch::Tensor amax_next);

// sampling.hip
std::vector<torch::Tensor> sample_logprob(torch::Tensor logits, double temperature,
                                          int64_t seed, int64_t step,
                                          c10::optional<torch::Tensor> step_tensor);
torch::Tensor gather_logprob(torch::Tensor logits, torch::Tensor tokens, double temperature);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "RMSNorm forward (bf16)");
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "RMSNorm backward (bf16)");
  m.def("rope_inplace", &rope_inplace, "Rotary embedding in-place (bf16)");
  m.def("swiglu_fwd", &swiglu_fwd, "SwiGLU forward (bf16)");
  m.def("swiglu_bwd", &swiglu_bwd, "SwiGLU backward (bf16)");
  m.def("lse_chunk_update", &lse_chunk_update, "Online LSE + target gather over a vocab chunk");
  m.def("ce_bwd_chunk", &ce_bwd_chunk, "Chunked logprob backward: dlogits");
  m.def("grpo_loss_fwd", &grpo_loss_fwd, "Fused GRPO ratio-clip + k3 KL loss (loss, dlp, clipfrac)");
  m.def("grad_sq_sum", &grad_sq_sum, "Sum of squared gradients (bf16 flat)");
  m.def("adamw_step", &adamw_step, "Fused AdamW over flat buffers");
  m.def("flash_prefill", &flash_prefill, "Flash causal GQA prefill (varlen, MFMA)");
  m.def("flash_train_fwd", &flash_train_fwd, "Training flash attention forward (O + LSE)");
  m.def("flash_train_bwd", &flash_train_bwd, "Training flash attention backward (dQ,dK,dV)");
  m.def("paged_decode", &paged_decode, "Paged decode attention (flash-decoding splits)");
  m.def("gather_cache", &gather_cache, "Gather K/V rows from pages into contiguous buffers");
  m.def("reshape_and_cache", &reshape_and_cache, "Scatter K/V into KV pages");
  m.def("qkv_rope_cache", &qkv_rope_cache, "Fused bias+rope+cache-write+q-extract");
  m.def("add_rmsnorm_", &add_rmsnorm_, "Fused residual add (in-place) + RMSNorm");
  m.def("skinny_gemm", &skinny_gemm, "Weight-streaming skinny-M GEMM (decode path)");
  m.def("skinny_gemm_v3", &skinny_gemm_v3, "Skinny GEMM v3: 64x64 wave tiles, double-buffered LDS, partials split-K");
  m.def("hbl_tune", &hbl_tune, "Sweep hipblaslt heuristic algos for a decode shape; cache winner");
  m.def("hbl_has", &hbl_has, "Is this (M,N,K) tuned?");
  m.def("hbl_mm", &hbl_mm, "Tuned hipblaslt GEMM (graph-capture safe)");
  m.def("hbl_fp8_tune", &hbl_fp8_tune, "Tune fp8(e4m3) hipblaslt GEMM for a shape");
  m.def("hbl_fp8_has", &hbl_fp8_has, "Is this shape fp8-tuned?");
  m.def("hbl_fp8_mm", &hbl_fp8_mm, "Tuned fp8 GEMM with device scale pointers");
  m.def("fp8_quant_delayed", &fp8_quant_delayed, "Fused delayed-scaling e4m3 quantization");
  m.def("fp8_scale_update", &fp8_scale_update, "Fold accumulated amax into the fp8 scale");
  m.def("add_rmsnorm_fp8_", &add_rmsnorm_fp8_,
        "Fused residual add + RMSNorm emitting e4m3 (delayed scaling)");
  m.def("swiglu_fp8", &swiglu_fp8, "SwiGLU emitting e4m3 (delayed scaling)");
  m.def("sample_logprob", &sample_logprob, "Fused gumbel-max sampling + logprob");
  m.def("gather_logprob", &gather_logprob, "Logprob of given tokens from logits");
}
