// Python bindings for the sat_amd CDNA4 kernel layer (sat_amd._C).
#include <torch/extension.h>
#include <vector>

at::Tensor dense_fwd(at::Tensor x, at::Tensor w, at::Tensor bias, int64_t act);
at::Tensor dense_fwd_drop(at::Tensor x, at::Tensor w, at::Tensor seed,
                          double p, int64_t salt);
std::vector<at::Tensor> dense_lstm_expand_fwd(
        at::Tensor xh, at::Tensor wl, at::Tensor bl, at::Tensor c_prev,
        at::Tensor pooled, at::Tensor table, at::Tensor ids,
        at::Tensor seed, at::Tensor expdrop, at::Tensor od_next,
        double fb, double p_lstm, double p_fc, int64_t s);
std::vector<at::Tensor> dexp_lstm_bwd(at::Tensor dexpd,
                                      at::Tensor d_out_carry,
                                      at::Tensor d_sth_carry,
                                      at::Tensor seed, at::Tensor gates,
                                      at::Tensor c_prev, at::Tensor dc,
                                      at::Tensor dgates_out,
                                      double p_fc, double p_lstm,
                                      int64_t s, int64_t D, int64_t E,
                                      double fb);
std::vector<at::Tensor> dense_dx_fuse(at::Tensor dgates, at::Tensor wl_t,
                                      at::Tensor dpool_dec,
                                      at::Tensor demb_dec,
                                      at::Tensor seed,
                                      at::Tensor demb_out,
                                      double p, int64_t salt,
                                      int64_t D, int64_t E, int64_t H);
at::Tensor conv3_fwd(at::Tensor input, at::Tensor weight, at::Tensor bias,
                     bool relu, bool emit_pad);
void bias_act_nhwc(at::Tensor y, at::Tensor bias, bool relu);
void scale_bias_act_nhwc(at::Tensor y, at::Tensor scale, at::Tensor shift,
                         bool relu);
at::Tensor maxpool2x2_nhwc(at::Tensor input);
at::Tensor conv_igemm_fwd(at::Tensor input, at::Tensor w_ohwi,
                          at::Tensor bias, bool relu);
at::Tensor pad1_nhwc(at::Tensor input);
at::Tensor conv_igemm_8p_fwd(at::Tensor padded, at::Tensor w_ohwi,
                             at::Tensor bias, int64_t Hh, int64_t Ww,
                             bool relu);
at::Tensor conv_igemm_glds64_fwd(at::Tensor padded, at::Tensor w_ohwi,
                                 at::Tensor bias, int64_t Hh, int64_t Ww,
                                 bool relu);
at::Tensor dense_8p_fwd(at::Tensor x, at::Tensor w, at::Tensor bias,
                        int64_t act);
at::Tensor conv3x3_wgrad(at::Tensor xpad, at::Tensor dy_rows,
                         int64_t Hh, int64_t Ww);
at::Tensor conv_igemm_glds_fwd(at::Tensor padded, at::Tensor w_ohwi,
                               at::Tensor bias, int64_t Hh, int64_t Ww,
                               bool relu);
std::vector<at::Tensor> dense_lstm_fwd(at::Tensor xh, at::Tensor wl,
                                       at::Tensor bl, at::Tensor c_prev,
                                       double fb);
void dense_drop_fwd(at::Tensor x, at::Tensor w, at::Tensor b, int64_t act,
                    at::Tensor seed, double p, int64_t salt,
                    at::Tensor y, at::Tensor ydrop);
at::Tensor dense_fwd_out(at::Tensor x, at::Tensor w, at::Tensor bias,
                         int64_t act, at::Tensor out);
std::vector<at::Tensor> lstm_pointwise_bwd_out(at::Tensor gates,
                                               at::Tensor c, at::Tensor dh,
                                               at::Tensor dc, double fb,
                                               at::Tensor dgates_out);
void lstm_in_fuse(at::Tensor pooled, at::Tensor table, at::Tensor ids,
                  at::Tensor sth, at::Tensor seed, double p, int64_t salt,
                  at::Tensor xh);
std::vector<at::Tensor> expand_fuse(at::Tensor h_raw, at::Tensor pooled,
                                    at::Tensor table, at::Tensor ids,
                                    at::Tensor seed,
                                    at::Tensor expdrop, at::Tensor od_next,
                                    double p_lstm, double p_fc, int64_t s);
std::vector<at::Tensor> dexp_fuse(at::Tensor dexpd, at::Tensor d_out_carry,
                                  at::Tensor d_sth_carry, at::Tensor seed,
                                  double p_fc, double p_lstm, int64_t s,
                                  int64_t D, int64_t E);
std::vector<at::Tensor> dx_fuse(at::Tensor dxh, at::Tensor dpool_dec,
                                at::Tensor demb_dec, at::Tensor seed,
                                at::Tensor demb_out, double p,
                                int64_t salt, int64_t H);
void hash_dropout_out(at::Tensor x, at::Tensor seed, double p,
                      int64_t salt, at::Tensor out);
at::Tensor hash_dropout_steps(at::Tensor x, at::Tensor seed, double p,
                              int64_t salt_base, int64_t salt_stride,
                              int64_t T);
void act_bwd_out(at::Tensor dy, at::Tensor y, int64_t act, at::Tensor out);
at::Tensor hash_dropout_slabs(at::Tensor x, at::Tensor seed, double p,
                              int64_t salt_base, int64_t salt_stride,
                              int64_t T);
std::vector<at::Tensor> lstm_pointwise_fwd(at::Tensor gates, at::Tensor c,
                                           double fb);
std::vector<at::Tensor> lstm_pointwise_bwd(at::Tensor gates, at::Tensor c,
                                           at::Tensor dh, at::Tensor dc,
                                           double fb);
at::Tensor act_bwd(at::Tensor dy, at::Tensor y, int64_t act);
void act_bwd_f32_out(at::Tensor dy, at::Tensor y, int64_t act,
                     at::Tensor out);
at::Tensor hash_dropout(at::Tensor x, at::Tensor seed, double p,
                        int64_t salt);
std::vector<at::Tensor> attn_pool_fwd(at::Tensor ctx, at::Tensor logits);
std::vector<at::Tensor> attn_scores_fused(at::Tensor t1, at::Tensor t2,
                                          at::Tensor v, at::Tensor seed,
                                          double p, int64_t salt, int64_t L);
std::vector<at::Tensor> attn_pool_bwd(at::Tensor ctx, at::Tensor alpha,
                                      at::Tensor dalpha, at::Tensor dpooled,
                                      bool need_dctx);
std::vector<at::Tensor> attn_scores_bwd(at::Tensor tdrop, at::Tensor v,
                                        at::Tensor dlogits, at::Tensor seed,
                                        double p, int64_t salt, int64_t L);
std::vector<at::Tensor> attn_scores_bwd_acc(at::Tensor tdrop, at::Tensor v,
                                            at::Tensor dlogits,
                                            at::Tensor seed, double p,
                                            int64_t salt, int64_t L,
                                            at::Tensor dv_acc);
std::vector<at::Tensor> attn_scores_bwd_tanh(
        at::Tensor tdrop, at::Tensor v, at::Tensor dlogits,
        at::Tensor seed, double p, int64_t salt, int64_t L,
        at::Tensor dv_acc, at::Tensor t1y, at::Tensor dt1_out);
at::Tensor embedding_fwd(at::Tensor ids, at::Tensor table);
at::Tensor embedding_bwd(at::Tensor ids, at::Tensor dy, int64_t rows);
std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor labels,
                               at::Tensor mask);
at::Tensor ce_bwd(at::Tensor logits, at::Tensor labels, at::Tensor mask,
                  at::Tensor lse, at::Tensor dloss);
at::Tensor grad_sq_norm(std::vector<at::Tensor> grads);
at::Tensor sq_norm_mt(at::Tensor desc, at::Tensor cum, int64_t n_tensors,
                      int64_t total);
void adam_step_mt(at::Tensor desc, at::Tensor cum, int64_t n_tensors,
                  int64_t total, at::Tensor step_dev, double lr0,
                  double decay_factor, double steps_per_decay, double b1,
                  double b2, double eps, double clip, at::Tensor gsq,
                  bool zero_grads, bool vec4);
void adam_step(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
               std::vector<at::Tensor> ms, std::vector<at::Tensor> vs,
               at::Tensor step_dev, double lr0, double decay_factor,
               double steps_per_decay, double b1, double b2, double eps,
               double clip, at::Tensor gsq);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
    m.def("dense_fwd", &dense_fwd, "MFMA GEMM + bias/act (bf16)");
    m.def("dense_fwd_drop", &dense_fwd_drop,
          "skinny GEMM with hash-dropout fused into the split-K epilogue");
    m.def("dense_lstm_expand_fwd", &dense_lstm_expand_fwd,
          "gates GEMM + LSTM gate math + expand scatter in 2 launches");
    m.def("dexp_lstm_bwd", &dexp_lstm_bwd,
          "dexp scatter + LSTM pointwise backward in one launch");
    m.def("dense_dx_fuse", &dense_dx_fuse,
          "dxh GEMM with the dx scatter fused into the epilogue");
    m.def("bias_act_nhwc", &bias_act_nhwc);
    m.def("scale_bias_act_nhwc", &scale_bias_act_nhwc);
    m.def("maxpool2x2_nhwc", &maxpool2x2_nhwc);
    m.def("conv_igemm_fwd", &conv_igemm_fwd);
    m.def("pad1_nhwc", &pad1_nhwc);
    m.def("conv_igemm_glds_fwd", &conv_igemm_glds_fwd);
    m.def("conv_igemm_8p_fwd", &conv_igemm_8p_fwd,
          "8-phase deep-pipelined implicit-GEMM conv (Cout%256==0)");
    m.def("conv_igemm_glds64_fwd", &conv_igemm_glds64_fwd);
    m.def("dense_8p_fwd", &dense_8p_fwd,
          "8-phase deep-pipelined dense GEMM (big-M shapes)");
    m.def("conv3x3_wgrad", &conv3x3_wgrad,
          "3x3 conv weight grad: transpose-staged MFMA split-K reduce");
    m.def("dense_lstm_fwd", &dense_lstm_fwd);
    m.def("dense_drop_fwd", &dense_drop_fwd);
    m.def("conv3_fwd", &conv3_fwd,
          "direct NHWC conv for 3-channel 3x3/s1 (VGG conv1_1)");
    m.def("dense_fwd_out", &dense_fwd_out);
    m.def("lstm_pointwise_bwd_out", &lstm_pointwise_bwd_out);
    m.def("lstm_in_fuse", &lstm_in_fuse);
    m.def("expand_fuse", &expand_fuse);
    m.def("dexp_fuse", &dexp_fuse);
    m.def("dx_fuse", &dx_fuse);
    m.def("hash_dropout_out", &hash_dropout_out);
    m.def("hash_dropout_steps", &hash_dropout_steps);
    m.def("act_bwd_out", &act_bwd_out);
    m.def("hash_dropout_slabs", &hash_dropout_slabs);
    m.def("lstm_pointwise_fwd", &lstm_pointwise_fwd);
    m.def("lstm_pointwise_bwd", &lstm_pointwise_bwd);
    m.def("act_bwd", &act_bwd);
    m.def("act_bwd_f32_out", &act_bwd_f32_out);
    m.def("hash_dropout", &hash_dropout);
    m.def("attn_pool_fwd", &attn_pool_fwd);
    m.def("attn_scores_fused", &attn_scores_fused);
    m.def("attn_pool_bwd", &attn_pool_bwd);
    m.def("attn_scores_bwd", &attn_scores_bwd);
    m.def("attn_scores_bwd_acc", &attn_scores_bwd_acc);
    m.def("attn_scores_bwd_tanh", &attn_scores_bwd_tanh,
          "scores backward with fused tanh-bwd dt1 into a given slab");
    m.def("embedding_fwd", &embedding_fwd);
    m.def("embedding_bwd", &embedding_bwd);
    m.def("ce_fwd", &ce_fwd);
    m.def("ce_bwd", &ce_bwd);
    m.def("grad_sq_norm", &grad_sq_norm);
    m.def("sq_norm_mt", &sq_norm_mt);
    m.def("adam_step_mt", &adam_step_mt);
    m.def("adam_step", &adam_step);
}
