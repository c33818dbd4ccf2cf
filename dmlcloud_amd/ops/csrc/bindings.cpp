// Python bindings for the dmlcloud_amd gfx950 kernels.
//
// This TU is compiled by the host compiler; all kernel launchers live in
// the .hip TUs and are declared here.

#include <torch/extension.h>

namespace dmlamd {

// reduce.hip
void metric_reduce_into(at::Tensor value, at::Tensor acc, at::Tensor count, at::Tensor partials,
                        int64_t op);
void metric_accumulate_elementwise(at::Tensor value, at::Tensor acc, at::Tensor count,
                                   int64_t op);
at::Tensor metric_finalize_dims(at::Tensor acc, std::vector<int64_t> dims, int64_t op);

// copy.hip
void chunked_copy(at::Tensor units_blob, int64_t nunits);

// optim.hip
void fused_adam(at::Tensor param, at::Tensor grad, at::Tensor exp_avg, at::Tensor exp_avg_sq,
                at::Tensor step_t, double lr, double beta1, double beta2, double eps,
                double weight_decay, double grad_scale);
void fused_sgd(at::Tensor param, at::Tensor grad, at::Tensor momentum_buf, double lr,
               double momentum, double weight_decay, double grad_scale, bool use_momentum);
void fused_adam_bf16(at::Tensor param, at::Tensor grad, at::Tensor master, at::Tensor exp_avg,
                     at::Tensor exp_avg_sq, at::Tensor step_t, double lr, double beta1,
                     double beta2, double eps, double weight_decay, double grad_scale);
void fused_sgd_bf16(at::Tensor param, at::Tensor grad, at::Tensor master,
                    at::Tensor momentum_buf, double lr, double momentum, double weight_decay,
                    double grad_scale, bool use_momentum);
void l2_norm_and_scale(at::Tensor flat, at::Tensor partials, at::Tensor out, double max_norm,
                       bool apply, double norm_scale);

// smallcnn.hip
void conv3x3_relu_pool_fwd(at::Tensor in, at::Tensor w, at::Tensor b, at::Tensor out,
                           at::Tensor argmax);
void conv3x3_relu_pool_bwd_data(at::Tensor dpooled, at::Tensor argmax, at::Tensor pooled,
                                at::Tensor w, at::Tensor din);
void conv3x3_relu_pool_bwd_weight(at::Tensor dpooled, at::Tensor argmax, at::Tensor pooled,
                                  at::Tensor in, at::Tensor dw, at::Tensor db);

// layernorm.hip
void layernorm_fwd(at::Tensor x, at::Tensor gamma, at::Tensor beta, at::Tensor y,
                   at::Tensor mean, at::Tensor rstd, double eps);
void layernorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor mean, at::Tensor rstd,
                   at::Tensor gamma, at::Tensor dx, at::Tensor dgb_ws, at::Tensor dgamma,
                   at::Tensor dbeta);

// loss.hip
void ce_fwd(at::Tensor logits, at::Tensor targets, at::Tensor loss, at::Tensor lse,
            int64_t ignore_index);
void ce_bwd(at::Tensor logits, at::Tensor targets, at::Tensor lse, at::Tensor scale,
            at::Tensor dlogits, int64_t ignore_index);

// attention.hip (experimental)
void attn_fwd(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor o, at::Tensor lse,
              double scale, bool causal);
void attn_bwd(at::Tensor q, at::Tensor k, at::Tensor v, at::Tensor dout, at::Tensor o,
              at::Tensor lse, at::Tensor dq, at::Tensor dk, at::Tensor dv, at::Tensor delta,
              double scale, bool causal);

} // namespace dmlamd

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "dmlcloud_amd gfx950 (MI355X/CDNA4) kernels";
  m.def("metric_reduce_into", &dmlamd::metric_reduce_into,
        "Fully reduce value into scalar accumulator (deterministic)");
  m.def("metric_accumulate_elementwise", &dmlamd::metric_accumulate_elementwise,
        "Elementwise merge value into accumulator");
  m.def("metric_finalize_dims", &dmlamd::metric_finalize_dims,
        "Reduce accumulator over given dims");
  m.def("chunked_copy", &dmlamd::chunked_copy, "Descriptor-table gather/scatter copy");
  m.def("fused_adam", &dmlamd::fused_adam, "Fused Adam on flat fp32 buffers");
  m.def("fused_sgd", &dmlamd::fused_sgd, "Fused SGD on flat fp32 buffers");
  m.def("fused_adam_bf16", &dmlamd::fused_adam_bf16,
        "Fused Adam: bf16 params/grads, fp32 master + moments");
  m.def("fused_sgd_bf16", &dmlamd::fused_sgd_bf16,
        "Fused SGD: bf16 params/grads, fp32 master");
  m.def("l2_norm_and_scale", &dmlamd::l2_norm_and_scale,
        "Deterministic L2 norm + optional clip scale");
  m.def("conv3x3_relu_pool_fwd", &dmlamd::conv3x3_relu_pool_fwd,
        "Fused conv3x3(pad1)+ReLU+maxpool2x2 forward");
  m.def("conv3x3_relu_pool_bwd_data", &dmlamd::conv3x3_relu_pool_bwd_data,
        "Fused conv+relu+pool input gradient");
  m.def("conv3x3_relu_pool_bwd_weight", &dmlamd::conv3x3_relu_pool_bwd_weight,
        "Fused conv+relu+pool weight/bias gradient");
  m.def("layernorm_fwd", &dmlamd::layernorm_fwd, "Fused bf16 LayerNorm forward (row per wave)");
  m.def("layernorm_bwd", &dmlamd::layernorm_bwd,
        "Fused bf16 LayerNorm backward (dx + dgamma/dbeta)");
  m.def("ce_fwd", &dmlamd::ce_fwd, "Online softmax cross-entropy forward (per-row loss + lse)");
  m.def("ce_bwd", &dmlamd::ce_bwd, "Cross-entropy backward (dlogits in one pass)");
  m.def("attn_fwd", &dmlamd::attn_fwd,
        "EXPERIMENTAL flash-attention forward (MFMA bf16, D=64, emits lse)");
  m.def("attn_bwd", &dmlamd::attn_bwd,
        "EXPERIMENTAL flash-attention backward (recompute; dq/dk/dv)");
}
