// Python bindings for the MI355X HIP kernel layer (_mcdp_C).
#include <torch/extension.h>

// rmsnorm.hip
std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps);
std::vector<at::Tensor> rmsnorm_bwd(at::Tensor x, at::Tensor w, at::Tensor rstd, at::Tensor dy);
std::vector<at::Tensor> rmsnorm_fwd_res(at::Tensor x, at::Tensor res, at::Tensor w, double eps, bool with_amax);
std::vector<at::Tensor> rmsnorm_bwd_add(at::Tensor x, at::Tensor w, at::Tensor rstd, at::Tensor dy, at::Tensor dadd);
// rope.hip
at::Tensor rope_fwd(at::Tensor x, at::Tensor cost, at::Tensor sint, bool traditional,
                    long offset, bool conj);
at::Tensor rope_fwd_out(at::Tensor x, at::Tensor cost, at::Tensor sint, bool traditional,
                        long offset, bool conj, at::Tensor y);
// swiglu.hip
at::Tensor swiglu_fwd(at::Tensor gu);
std::vector<at::Tensor> swiglu_fwd_amax(at::Tensor gu, bool with_amax);
at::Tensor swiglu_bwd(at::Tensor gu, at::Tensor dy);
// cross_entropy.hip
std::vector<at::Tensor> ce_vp_stats(at::Tensor logits, at::Tensor targets, long v0, long ignore_index);
at::Tensor ce_vp_sumexp(at::Tensor logits, at::Tensor m_group);
at::Tensor ce_vp_bwd(at::Tensor logits, at::Tensor targets, at::Tensor lse, at::Tensor scale, long v0, long ignore_index);
std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor targets, long ignore_index);
at::Tensor ce_bwd(at::Tensor logits, at::Tensor targets, at::Tensor lse, at::Tensor scale,
                  long ignore_index);
// optim.hip
at::Tensor sumsq(at::Tensor g);
void adamw_step(at::Tensor param, at::Tensor master, at::Tensor grad, at::Tensor m,
                at::Tensor v, at::Tensor sumsq_t, long step, double lr, double b1,
                double b2, double eps, double wd, long decay_boundary, double max_norm);
void lion_step(at::Tensor param, at::Tensor master, at::Tensor grad, at::Tensor m,
               at::Tensor sumsq_t, double lr, double b1, double b2, double wd,
               long decay_boundary, double max_norm);
void sgd_step(at::Tensor param, at::Tensor master, at::Tensor grad, at::Tensor buf,
              at::Tensor sumsq_t, double lr, double mom, double wd, long decay_boundary,
              bool nesterov, double max_norm);
// muon.hip (K8 Newton-Schulz GEMM chain)
void muon_gemm_nt(at::Tensor X, at::Tensor Y, at::Tensor C, double alpha,
                  double beta, at::Tensor E);
void muon_gemm_nn_ax(at::Tensor Bm, at::Tensor X, at::Tensor C, double a);
void shampoo_stats_update(at::Tensor G, at::Tensor S, double beta);

// attn_fwd.hip / attn_bwd.hip
std::vector<at::Tensor> attn_fwd_blockmask(at::Tensor q, at::Tensor k, at::Tensor v,
                                           double scale, at::Tensor gran, at::Tensor bits,
                                           at::Tensor range, at::Tensor bias);
std::vector<at::Tensor> attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v, double scale,
                                 long mod, long modarg, at::Tensor slopes);
std::vector<at::Tensor> attn_bwd(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor o,
                                 at::Tensor dout, at::Tensor lse, double scale, long mod,
                                 long modarg, at::Tensor slopes);
std::vector<at::Tensor> attn_bwd_out(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor o,
                                     at::Tensor dout, at::Tensor lse, double scale, long mod,
                                     long modarg, at::Tensor slopes,
                                     at::Tensor dq, at::Tensor dk, at::Tensor dv);
// sampling.hip
at::Tensor sample_token(at::Tensor logits, double temperature, double top_p, double min_p,
                        long seed);
void sample_token_dev(at::Tensor logits, at::Tensor out, double temperature,
                      double min_p, long seed, at::Tensor pos);
// decode.hip (hipGraph-capturable static decode)
void rope_decode_(at::Tensor x, at::Tensor cost, at::Tensor sint, bool traditional,
                  at::Tensor pos);
void kv_append_(at::Tensor k, at::Tensor v, at::Tensor kc, at::Tensor vc, at::Tensor pos);
at::Tensor attn_decode(at::Tensor q, at::Tensor kc, at::Tensor vc, at::Tensor pos,
                       at::Tensor part, double scale);
void pos_incr_(at::Tensor pos);
void write_token_(at::Tensor tok, at::Tensor ring, at::Tensor idx);
void kv_append_q8_(at::Tensor k, at::Tensor v, at::Tensor kc, at::Tensor ksz,
                   at::Tensor vc, at::Tensor vsz, at::Tensor pos);
at::Tensor attn_decode_q8(at::Tensor q, at::Tensor kc, at::Tensor ksz, at::Tensor vc,
                          at::Tensor vsz, at::Tensor pos, at::Tensor part, double scale);
// gemv.hip
at::Tensor gemv_bf16(at::Tensor x, at::Tensor W);
at::Tensor gemv_ex(at::Tensor x, at::Tensor W, long mode, at::Tensor nw, double eps,
                   at::Tensor res);
// fp8_quant.hip
std::vector<at::Tensor> fp8_quantize(at::Tensor x, bool transpose);
std::vector<at::Tensor> fp8_quantize_pre(at::Tensor x, at::Tensor amax_in, bool transpose);
// debug.hip
at::Tensor mfma_tile_test(at::Tensor A, at::Tensor B);
at::Tensor afrag_transform_test(at::Tensor M);
at::Tensor tr16_frag_test(at::Tensor X);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("rmsnorm_fwd", &rmsnorm_fwd, "RMSNorm forward (y, rstd)");
  m.def("rmsnorm_bwd", &rmsnorm_bwd, "RMSNorm backward (dx, dw)");
  m.def("rmsnorm_fwd_res", &rmsnorm_fwd_res, "fused residual-add + RMSNorm fwd (y, rstd[, s][, amax])");
  m.def("rmsnorm_bwd_add", &rmsnorm_bwd_add, "RMSNorm bwd with fused grad add (dx, dw)");
  m.def("rope_fwd", &rope_fwd, "RoPE apply (conj=true for backward)");
  m.def("rope_fwd_out", &rope_fwd_out, "RoPE apply into a strided out view");
  m.def("swiglu_fwd", &swiglu_fwd, "SwiGLU forward");
  m.def("swiglu_fwd_amax", &swiglu_fwd_amax, "swiglu fwd emitting |out| amax bits");
  m.def("swiglu_bwd", &swiglu_bwd, "SwiGLU backward");
  m.def("ce_fwd", &ce_fwd, "fused cross-entropy forward (loss_sum, ntok, lse)");
  m.def("ce_vp_stats", &ce_vp_stats, "vocab-parallel CE pass 1 (m_local, tgt_local)");
  m.def("ce_vp_sumexp", &ce_vp_sumexp, "vocab-parallel CE pass 2 (sum exp)");
  m.def("ce_vp_bwd", &ce_vp_bwd, "vocab-parallel CE backward (dlogits)");
  m.def("ce_bwd", &ce_bwd, "fused cross-entropy backward (dlogits)");
  m.def("sumsq", &sumsq, "sum of squares -> f32 scalar");
  m.def("adamw_step", &adamw_step, "fused AdamW over flat buffers");
  m.def("lion_step", &lion_step, "fused Lion over flat buffers");
  m.def("sgd_step", &sgd_step, "fused SGD over flat buffers");
  m.def("muon_gemm_nt", &muon_gemm_nt, "C = alpha*X@Y^T + beta*E (bf16 MFMA)");
  m.def("muon_gemm_nn_ax", &muon_gemm_nn_ax, "C = Bm@X + a*X (bf16 MFMA)");
  m.def("shampoo_stats_update", &shampoo_stats_update,
        "S = beta*S + (1-beta)*G@G^T (fp32 state, bf16 MFMA)");
  m.def("attn_fwd", &attn_fwd, "flash attention forward (o, lse)");
  m.def("attn_fwd_blockmask", &attn_fwd_blockmask,
        "flash attention fwd with device block-mask/bits/bias (K2 flex)");
  m.def("attn_bwd", &attn_bwd, "flash attention backward (dq, dk, dv)");
  m.def("attn_bwd_out", &attn_bwd_out,
        "flash attention backward into strided out views (fused dQKV)");
  m.def("sample_token", &sample_token, "fused temperature/min-p sampling");
  m.def("sample_token_dev", &sample_token_dev,
        "graph-capturable sampling (device position salt)");
  m.def("rope_decode_", &rope_decode_, "in-place RoPE at device position");
  m.def("kv_append_", &kv_append_, "append k/v into the static cache at device position");
  m.def("attn_decode", &attn_decode, "split-KV decode attention over the static cache");
  m.def("pos_incr_", &pos_incr_, "device position += 1");
  m.def("write_token_", &write_token_, "record token into the device ring");
  m.def("kv_append_q8_", &kv_append_q8_, "quantize+append k/v into the int8 static cache");
  m.def("attn_decode_q8", &attn_decode_q8, "decode attention over the int8 cache (fused dequant)");
  m.def("gemv_bf16", &gemv_bf16, "bf16 GEMV (decode projections)");
  m.def("gemv_ex", &gemv_ex,
        "fused decode GEMV (rmsnorm/swiglu staging, residual epilogue)");
  m.def("fp8_quantize", &fp8_quantize, "fused bf16 -> e4m3 quantize (codes, scale)");
  m.def("fp8_quantize_pre", &fp8_quantize_pre, "e4m3 quantize with producer-supplied per-block amax partials");
  m.def("mfma_tile_test", &mfma_tile_test, "debug: one 32x32x16 MFMA tile");
  m.def("afrag_transform_test", &afrag_transform_test, "debug: acc->A-frag transform");
  m.def("tr16_frag_test", &tr16_frag_test, "debug: ds_read_b64_tr_b16 B-fragment gather");
}
