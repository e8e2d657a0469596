// Python bindings for the trlx_amd gfx950 kernels (trlx_amd._C).
#include <torch/extension.h>

std::vector<at::Tensor> logprobs_fwd(const at::Tensor& logits, const at::Tensor& labels);
at::Tensor logprobs_bwd(const at::Tensor& logits, const at::Tensor& labels, const at::Tensor& lse,
                        const at::Tensor& gout);
std::vector<at::Tensor> rmsnorm_fwd(const at::Tensor& x, const at::Tensor& w, double eps,
                                    const c10::optional<at::Tensor>& res);
std::vector<at::Tensor> rmsnorm_bwd(const at::Tensor& x, const at::Tensor& w,
                                    const at::Tensor& invr, const at::Tensor& dy,
                                    const c10::optional<at::Tensor>& dres);
std::vector<at::Tensor> layernorm_fwd(const at::Tensor& x, const at::Tensor& w,
                                      const c10::optional<at::Tensor>& b, double eps,
                                      const c10::optional<at::Tensor>& res);
std::vector<at::Tensor> layernorm_bwd(const at::Tensor& x, const at::Tensor& w,
                                      const at::Tensor& mean, const at::Tensor& invstd,
                                      const at::Tensor& dy,
                                      const c10::optional<at::Tensor>& dres);
at::Tensor rope_fwd(const at::Tensor& x, const at::Tensor& cos, const at::Tensor& sin,
                    const at::Tensor& pos, bool interleaved, bool inverse, long rot);
std::vector<at::Tensor> gae(const at::Tensor& values, const at::Tensor& rewards, double gamma,
                            double lam);
at::Tensor sum_count(const at::Tensor& x);
at::Tensor normalize(const at::Tensor& x, const at::Tensor& mean, const at::Tensor& var,
                     bool shift_mean);
at::Tensor gumbel_sample(const at::Tensor& logits, double temperature,
                         const c10::optional<at::Tensor>& thresholds, long seed, long offset);
at::Tensor gumbel_sample_dev(const at::Tensor& logits, double temperature,
                             const c10::optional<at::Tensor>& thresholds, long seed,
                             const at::Tensor& offset);
at::Tensor decode_prep(const at::Tensor& qkv, at::Tensor& kcache, at::Tensor& vcache,
                       const c10::optional<at::Tensor>& cos, const c10::optional<at::Tensor>& sin,
                       const c10::optional<at::Tensor>& key_starts, const at::Tensor& cache_idx,
                       long num_heads, long rot, bool interleaved);
std::vector<at::Tensor> qkv_prep_fwd(const at::Tensor& qkv, long num_heads, long num_kv_heads,
                                     long head_dim, const c10::optional<at::Tensor>& cos,
                                     const c10::optional<at::Tensor>& sin,
                                     const c10::optional<at::Tensor>& pos, double qscale,
                                     long rot, bool interleaved);
at::Tensor qkv_prep_bwd(const at::Tensor& dq, const at::Tensor& dk, const at::Tensor& dv,
                        const c10::optional<at::Tensor>& cos, const c10::optional<at::Tensor>& sin,
                        const c10::optional<at::Tensor>& pos, double qscale, long rot,
                        bool interleaved);
at::Tensor causal_softmax_fwd(const at::Tensor& scores, long start_pos,
                              const c10::optional<at::Tensor>& key_starts);
at::Tensor causal_softmax_bwd(const at::Tensor& probs, const at::Tensor& dprobs);
at::Tensor attention_decode(const at::Tensor& q, const at::Tensor& kc, const at::Tensor& vc,
                            const at::Tensor& seq_lens, double scale,
                            const c10::optional<at::Tensor>& seq_starts);
at::Tensor lm_logprobs_v2(const at::Tensor& hidden, const at::Tensor& weight,
                          const at::Tensor& labels);
at::Tensor skinny_gemm(const at::Tensor& a, const at::Tensor& w,
                       const c10::optional<at::Tensor>& bias, long act);
at::Tensor skinny_gemm_fp8(const at::Tensor& a, const at::Tensor& w8, const at::Tensor& wscale,
                           const c10::optional<at::Tensor>& bias, long act);
std::vector<at::Tensor> lm_logprobs_v2_with_lse(const at::Tensor& hidden,
                                                const at::Tensor& weight,
                                                const at::Tensor& labels, bool want_lse);
at::Tensor ce_dlogits(const at::Tensor& hidden, const at::Tensor& weight,
                      const at::Tensor& labels, const at::Tensor& lse, const at::Tensor& dlp);
at::Tensor fused_decode_attention(const at::Tensor& qkv, at::Tensor& kcache, at::Tensor& vcache,
                                  const at::Tensor& seq_lens,
                                  const c10::optional<at::Tensor>& seq_starts,
                                  const c10::optional<at::Tensor>& cos,
                                  const c10::optional<at::Tensor>& sin,
                                  const at::Tensor& cache_idx, long rot, bool interleaved,
                                  double scale);
void decode_advance(const at::Tensor& tok, at::Tensor& out_tokens, at::Tensor& cur_tok,
                    at::Tensor& finished, at::Tensor& rng_offset, at::Tensor& step_col,
                    at::Tensor& cache_idx, at::Tensor& seq_lens, at::Tensor& pos_ids,
                    const c10::optional<at::Tensor>& key_starts, long eos, long pad);
at::Tensor lm_logprobs(const at::Tensor& hidden, const at::Tensor& weight,
                       const at::Tensor& labels);
void fused_adamw(at::Tensor& p, at::Tensor& master, const at::Tensor& g, at::Tensor& m,
                 at::Tensor& v, long step, double lr, double beta1, double beta2, double eps,
                 double weight_decay, double grad_scale);
void stage_gemm(const at::Tensor& a, const at::Tensor& w, const c10::optional<at::Tensor>& bias,
                at::Tensor& c, const c10::optional<at::Tensor>& nstats,
                const c10::optional<at::Tensor>& nw, const c10::optional<at::Tensor>& nb,
                bool norm_rms, double eps, long act, const c10::optional<at::Tensor>& resid,
                const c10::optional<at::Tensor>& out_stats);
void embed_stats(const at::Tensor& wte, const c10::optional<at::Tensor>& wpe,
                 const at::Tensor& cur_tok, const at::Tensor& pos_ids, long pos_offset,
                 at::Tensor& x, at::Tensor& stats, at::Tensor& packed);
void lm_sample(const at::Tensor& x, const at::Tensor& wlm, const c10::optional<at::Tensor>& blm,
               const at::Tensor& nstats, const at::Tensor& nw,
               const c10::optional<at::Tensor>& nb, at::Tensor& packed, bool norm_rms,
               double eps, double temperature, long seed, const at::Tensor& rng_offset);
void advance_packed(const at::Tensor& packed, at::Tensor& out_tokens, at::Tensor& cur_tok,
                    at::Tensor& finished, at::Tensor& rng_offset, at::Tensor& step_col,
                    at::Tensor& cache_idx, at::Tensor& seq_lens, at::Tensor& pos_ids,
                    const c10::optional<at::Tensor>& key_starts, long eos, long pad);
void stage_gemm_v2(const at::Tensor& a, const at::Tensor& w, const c10::optional<at::Tensor>& bias,
                   at::Tensor& c, bool norm, const c10::optional<at::Tensor>& nw,
                   const c10::optional<at::Tensor>& nb, bool norm_rms, double eps, long act,
                   const c10::optional<at::Tensor>& resid,
                   const c10::optional<at::Tensor>& pstats_out);
void stage_gemm_v3(const at::Tensor& a, const at::Tensor& w, const c10::optional<at::Tensor>& bias,
                   at::Tensor& c, const c10::optional<at::Tensor>& pstats, long nparts,
                   const c10::optional<at::Tensor>& nw, const c10::optional<at::Tensor>& nb,
                   bool norm_rms, double eps, long act,
                   const c10::optional<at::Tensor>& pstats_out);
at::Tensor flash_prefill(const at::Tensor& q, const at::Tensor& k, const at::Tensor& v,
                         const c10::optional<at::Tensor>& key_starts, long start_pos,
                         double scale, long tk);
std::vector<at::Tensor> flash_prefill_lse(const at::Tensor& q, const at::Tensor& k,
                                          const at::Tensor& v,
                                          const c10::optional<at::Tensor>& key_starts,
                                          long start_pos, double scale, long tk,
                                          bool want_lse);
std::vector<at::Tensor> flash_prefill_bwd(const at::Tensor& q, const at::Tensor& k,
                                          const at::Tensor& v, const at::Tensor& out,
                                          const at::Tensor& dout, const at::Tensor& lse,
                                          const c10::optional<at::Tensor>& key_starts,
                                          double scale);
void lm_sample_v3(const at::Tensor& x, const at::Tensor& wlm,
                  const c10::optional<at::Tensor>& blm, const at::Tensor& pstats, long nparts,
                  const at::Tensor& nw, const c10::optional<at::Tensor>& nb, at::Tensor& packed,
                  bool norm_rms, double eps, double temperature, long seed,
                  const at::Tensor& rng_offset);
void lm_sample_v2(const at::Tensor& x, const at::Tensor& wlm,
                  const c10::optional<at::Tensor>& blm, const at::Tensor& nw,
                  const c10::optional<at::Tensor>& nb, at::Tensor& packed, bool norm_rms,
                  double eps, double temperature, long seed, const at::Tensor& rng_offset);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, mod) {
  mod.def("logprobs_fwd", &logprobs_fwd, "fused logsumexp + label gather fwd");
  mod.def("logprobs_bwd", &logprobs_bwd, "logprobs backward");
  mod.def("rmsnorm_fwd", &rmsnorm_fwd);
  mod.def("rmsnorm_bwd", &rmsnorm_bwd);
  mod.def("layernorm_fwd", &layernorm_fwd);
  mod.def("layernorm_bwd", &layernorm_bwd);
  mod.def("rope_fwd", &rope_fwd);
  mod.def("gae", &gae);
  mod.def("sum_count", &sum_count);
  mod.def("normalize", &normalize);
  mod.def("gumbel_sample", &gumbel_sample);
  mod.def("gumbel_sample_dev", &gumbel_sample_dev);
  mod.def("decode_prep", &decode_prep);
  mod.def("qkv_prep_fwd", &qkv_prep_fwd);
  mod.def("qkv_prep_bwd", &qkv_prep_bwd);
  mod.def("causal_softmax_fwd", &causal_softmax_fwd);
  mod.def("causal_softmax_bwd", &causal_softmax_bwd);
  mod.def("attention_decode", &attention_decode);
  mod.def("fused_adamw", &fused_adamw);
  mod.def("lm_logprobs", &lm_logprobs);
  mod.def("lm_logprobs_v2", &lm_logprobs_v2);
  mod.def("skinny_gemm", &skinny_gemm);
  mod.def("skinny_gemm_fp8", &skinny_gemm_fp8);
  mod.def("lm_logprobs_v2_with_lse", &lm_logprobs_v2_with_lse);
  mod.def("ce_dlogits", &ce_dlogits);
  mod.def("decode_advance", &decode_advance);
  mod.def("fused_decode_attention", &fused_decode_attention);
  mod.def("stage_gemm", &stage_gemm);
  mod.def("embed_stats", &embed_stats);
  mod.def("lm_sample", &lm_sample);
  mod.def("advance_packed", &advance_packed);
  mod.def("stage_gemm_v2", &stage_gemm_v2);
  mod.def("lm_sample_v2", &lm_sample_v2);
  mod.def("stage_gemm_v3", &stage_gemm_v3);
  mod.def("lm_sample_v3", &lm_sample_v3);
  mod.def("flash_prefill", &flash_prefill);
  mod.def("flash_prefill_lse", &flash_prefill_lse);
  mod.def("flash_prefill_bwd", &flash_prefill_bwd);
}
