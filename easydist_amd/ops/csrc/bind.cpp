// torch library bindings for the gfx950 kernel extension.
#include <torch/extension.h>

std::tuple<at::Tensor, at::Tensor, at::Tensor>
layer_norm_fwd(const at::Tensor& x, const std::optional<at::Tensor>& w,
               const std::optional<at::Tensor>& b, double eps);
std::tuple<at::Tensor, at::Tensor, at::Tensor>
layer_norm_bwd(const at::Tensor& grad, const at::Tensor& x,
               const at::Tensor& mean, const at::Tensor& rstd,
               const std::optional<at::Tensor>& w, std::vector<bool> mask);
std::tuple<at::Tensor, at::Tensor>
rms_norm_fwd(const at::Tensor& x, const std::optional<at::Tensor>& w,
             double eps);
std::tuple<at::Tensor, at::Tensor>
rms_norm_bwd(const at::Tensor& grad, const at::Tensor& x,
             const at::Tensor& rstd, const std::optional<at::Tensor>& w);
std::tuple<at::Tensor, at::Tensor> ce_fwd(const at::Tensor& logits,
                                          const at::Tensor& targets);
std::tuple<at::Tensor, at::Tensor> ce_fwd_rows(const at::Tensor& logits,
                                               const at::Tensor& targets);
at::Tensor ce_bwd(const at::Tensor& grad, const at::Tensor& logits,
                  const at::Tensor& targets, const at::Tensor& lse);
std::tuple<std::vector<at::Tensor>, std::vector<at::Tensor>,
           std::vector<at::Tensor>, std::vector<at::Tensor>>
fused_adam_step(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
                std::vector<at::Tensor> exp_avgs,
                std::vector<at::Tensor> exp_avg_sqs,
                std::vector<at::Tensor> steps, double lr, double beta1,
                double beta2, double weight_decay, double eps);
std::tuple<std::vector<at::Tensor>, std::vector<at::Tensor>>
fused_sgd_step(std::vector<at::Tensor> params, std::vector<at::Tensor> grads,
               std::vector<at::Tensor> bufs, double lr, double momentum,
               double dampening, double weight_decay, bool nesterov);
at::Tensor gemm_nt(const at::Tensor& a, const at::Tensor& bt,
                   const std::optional<at::Tensor>& bias);
at::Tensor gemm_tn(const at::Tensor& a, const at::Tensor& b);
at::Tensor gemm_nt_act(const at::Tensor& a, const at::Tensor& bt,
                       const std::optional<at::Tensor>& bias, int64_t act,
                       const std::optional<at::Tensor>& aux);
std::tuple<at::Tensor, at::Tensor> gemm_tn_asum(const at::Tensor& a,
                                                const at::Tensor& b);
std::tuple<at::Tensor, at::Tensor>
gemm_nt_gelu(const at::Tensor& a, const at::Tensor& bt,
             const std::optional<at::Tensor>& bias, bool tanh_approx);
at::Tensor gelu_fast(const at::Tensor& x);
at::Tensor gelu_bwd_fast(const at::Tensor& g, const at::Tensor& x);
std::tuple<at::Tensor, at::Tensor> flash_attn_fwd(const at::Tensor& q,
                                                  const at::Tensor& k,
                                                  const at::Tensor& v,
                                                  bool causal);
std::tuple<at::Tensor, at::Tensor, at::Tensor>
flash_attn_bwd(const at::Tensor& grad, const at::Tensor& q,
               const at::Tensor& k, const at::Tensor& v,
               const at::Tensor& out, const at::Tensor& lse, bool causal);
at::Tensor
flash_attn_bwd_pack(const at::Tensor& grad, const at::Tensor& q,
                    const at::Tensor& k, const at::Tensor& v,
                    const at::Tensor& out, const at::Tensor& lse,
                    bool causal);

// wrappers adapting std::vector<bool> / TensorList signatures
static std::tuple<at::Tensor, at::Tensor, at::Tensor>
ln_bwd_wrap(const at::Tensor& grad, const at::Tensor& x,
            const at::Tensor& mean, const at::Tensor& rstd,
            const std::optional<at::Tensor>& w,
            std::vector<int64_t> mask) {
  std::vector<bool> m(mask.begin(), mask.end());
  return layer_norm_bwd(grad, x, mean, rstd, w, m);
}

static std::tuple<std::vector<at::Tensor>, std::vector<at::Tensor>,
                  std::vector<at::Tensor>, std::vector<at::Tensor>>
adam_wrap(at::TensorList params, at::TensorList grads, at::TensorList exp_avgs,
          at::TensorList exp_avg_sqs, at::TensorList steps, double lr,
          double beta1, double beta2, double weight_decay, double eps) {
  return fused_adam_step(params.vec(), grads.vec(), exp_avgs.vec(),
                         exp_avg_sqs.vec(), steps.vec(), lr, beta1, beta2,
                         weight_decay, eps);
}

static std::tuple<std::vector<at::Tensor>, std::vector<at::Tensor>>
sgd_wrap(at::TensorList params, at::TensorList grads, at::TensorList bufs,
         double lr, double momentum, double dampening, double weight_decay,
         bool nesterov) {
  return fused_sgd_step(params.vec(), grads.vec(), bufs.vec(), lr, momentum,
                        dampening, weight_decay, nesterov);
}

TORCH_LIBRARY(easydist_amd_hip, m) {
  m.def("layer_norm_fwd(Tensor x, Tensor? w, Tensor? b, float eps) "
        "-> (Tensor, Tensor, Tensor)");
  m.def("layer_norm_bwd(Tensor grad, Tensor x, Tensor mean, Tensor rstd, "
        "Tensor? w, int[] mask) -> (Tensor, Tensor, Tensor)");
  m.def("rms_norm_fwd(Tensor x, Tensor? w, float eps) -> (Tensor, Tensor)");
  m.def("rms_norm_bwd(Tensor grad, Tensor x, Tensor rstd, Tensor? w) "
        "-> (Tensor, Tensor)");
  m.def("ce_fwd(Tensor logits, Tensor targets) -> (Tensor, Tensor)");
  m.def("ce_fwd_rows(Tensor logits, Tensor targets) -> (Tensor, Tensor)");
  m.def("ce_bwd(Tensor grad, Tensor logits, Tensor targets, Tensor lse) "
        "-> Tensor");
  m.def("fused_adam_step(Tensor[] params, Tensor[] grads, Tensor[] exp_avgs, "
        "Tensor[] exp_avg_sqs, Tensor[] steps, float lr, float beta1, "
        "float beta2, float weight_decay, float eps) "
        "-> (Tensor[], Tensor[], Tensor[], Tensor[])");
  m.def("fused_sgd_step(Tensor[] params, Tensor[] grads, Tensor[] bufs, "
        "float lr, float momentum, float dampening, float weight_decay, "
        "bool nesterov) -> (Tensor[], Tensor[])");
  m.def("gemm_nt(Tensor a, Tensor bt, Tensor? bias) -> Tensor");
  m.def("gemm_tn(Tensor a, Tensor b) -> Tensor");
  m.def("gemm_nt_act(Tensor a, Tensor bt, Tensor? bias, int act, "
        "Tensor? aux) -> Tensor");
  m.def("gemm_tn_asum(Tensor a, Tensor b) -> (Tensor, Tensor)");
  m.def("gemm_nt_gelu(Tensor a, Tensor bt, Tensor? bias, bool tanh_approx) "
        "-> (Tensor, Tensor)");
  m.def("gelu_fast(Tensor x) -> Tensor");
  m.def("gelu_bwd_fast(Tensor g, Tensor x) -> Tensor");
  m.def("flash_attn_fwd(Tensor q, Tensor k, Tensor v, bool causal) "
        "-> (Tensor, Tensor)");
  m.def("flash_attn_bwd(Tensor grad, Tensor q, Tensor k, Tensor v, "
        "Tensor out, Tensor lse, bool causal) -> (Tensor, Tensor, Tensor)");
  m.def("flash_attn_bwd_pack(Tensor grad, Tensor q, Tensor k, Tensor v, "
        "Tensor out, Tensor lse, bool causal) -> Tensor");
}

TORCH_LIBRARY_IMPL(easydist_amd_hip, CUDA, m) {
  m.impl("layer_norm_fwd", layer_norm_fwd);
  m.impl("layer_norm_bwd", ln_bwd_wrap);
  m.impl("rms_norm_fwd", rms_norm_fwd);
  m.impl("rms_norm_bwd", rms_norm_bwd);
  m.impl("ce_fwd", ce_fwd);
  m.impl("ce_fwd_rows", ce_fwd_rows);
  m.impl("ce_bwd", ce_bwd);
  m.impl("fused_adam_step", adam_wrap);
  m.impl("fused_sgd_step", sgd_wrap);
  m.impl("gemm_nt", gemm_nt);
  m.impl("gemm_tn", gemm_tn);
  m.impl("gemm_nt_act", gemm_nt_act);
  m.impl("gemm_tn_asum", gemm_tn_asum);
  m.impl("gemm_nt_gelu", gemm_nt_gelu);
  m.impl("gelu_fast", gelu_fast);
  m.impl("gelu_bwd_fast", gelu_bwd_fast);
  m.impl("flash_attn_fwd", flash_attn_fwd);
  m.impl("flash_attn_bwd", flash_attn_bwd);
  m.impl("flash_attn_bwd_pack", flash_attn_bwd_pack);
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {}
