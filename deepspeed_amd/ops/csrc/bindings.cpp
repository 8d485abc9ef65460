// pybind bindings for the deepspeed_amd gfx950 kernel set.
#include <torch/extension.h>

void multi_tensor_adam(std::vector<at::Tensor> params,
                       std::vector<at::Tensor> grads,
                       std::vector<at::Tensor> exp_avgs,
                       std::vector<at::Tensor> exp_avg_sqs, double lr,
                       double beta1, double beta2, double eps, long step,
                       long adamw_mode, long bias_correction,
                       double weight_decay, std::vector<at::Tensor> out16,
                       double grad_scale);
void accum_bf16_to_f32(at::Tensor dst, at::Tensor src, double scale);
at::Tensor l2norm_sq(std::vector<at::Tensor> tensors);
at::Tensor fp8_amax(at::Tensor x);
at::Tensor ragged_decode(at::Tensor q, at::Tensor kpool, at::Tensor vpool,
                         at::Tensor rows, at::Tensor lens, long chunk);
at::Tensor fp8_cast(at::Tensor x, at::Tensor scale, bool e5m2);
std::vector<at::Tensor> fp8_cast_transpose(at::Tensor x, at::Tensor scale,
                                           bool e5m2);
std::vector<at::Tensor> flash_attn_fwd(at::Tensor q, at::Tensor k,
                                       at::Tensor v, bool causal,
                                       double scale,
                                       c10::optional<at::Tensor> kvmask);
std::vector<at::Tensor> flash_attn_bwd(at::Tensor q, at::Tensor k,
                                       at::Tensor v, at::Tensor dout,
                                       at::Tensor out, at::Tensor lse,
                                       bool causal, double scale,
                                       c10::optional<at::Tensor> kvmask);
void multi_tensor_adagrad(std::vector<at::Tensor> params,
                          std::vector<at::Tensor> grads,
                          std::vector<at::Tensor> sq_accums, double lr,
                          double eps, double weight_decay,
                          std::vector<at::Tensor> out16);
void multi_tensor_lamb(std::vector<at::Tensor> params,
                       std::vector<at::Tensor> grads,
                       std::vector<at::Tensor> updates,
                       std::vector<at::Tensor> exp_avgs,
                       std::vector<at::Tensor> exp_avg_sqs,
                       at::Tensor workspace, double lr, double beta1,
                       double beta2, double eps, long step,
                       long bias_correction, double weight_decay,
                       std::vector<at::Tensor> out16, double grad_scale);
void cpu_lion_step(at::Tensor p, at::Tensor g, at::Tensor m,
                   c10::optional<at::Tensor> out16, double lr, double beta1,
                   double beta2, double wd);
void cpu_adagrad_step(at::Tensor p, at::Tensor g, at::Tensor h,
                      c10::optional<at::Tensor> out16, double lr, double eps,
                      double wd);
void cpu_adam_step(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
                   c10::optional<at::Tensor> out16, double lr, double beta1,
                   double beta2, double eps, long step, long adamw,
                   long bias_correction, double wd, double grad_scale);
void multi_tensor_lion(std::vector<at::Tensor> params,
                       std::vector<at::Tensor> grads,
                       std::vector<at::Tensor> exp_avgs, double lr,
                       double beta1, double beta2, double weight_decay,
                       std::vector<at::Tensor> out16);
std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps);
std::vector<at::Tensor> rmsnorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                    at::Tensor rstd);
std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor w,
                                      c10::optional<at::Tensor> b, double eps);
std::vector<at::Tensor> layernorm_bwd(at::Tensor dy, at::Tensor x,
                                      at::Tensor w, at::Tensor mean,
                                      at::Tensor rstd);
void rope(at::Tensor dst, at::Tensor src, at::Tensor cos, at::Tensor sin,
          long pos0, bool backward);
void rope_inplace(at::Tensor t, at::Tensor cos, at::Tensor sin, long pos0,
                  bool backward);
at::Tensor swiglu_fwd(at::Tensor g, at::Tensor u);
std::vector<at::Tensor> swiglu_bwd(at::Tensor dy, at::Tensor g, at::Tensor u);
std::vector<at::Tensor> cross_entropy_fwd(at::Tensor logits,
                                          at::Tensor targets,
                                          long ignore_index);
at::Tensor cross_entropy_bwd(at::Tensor logits, at::Tensor targets,
                             at::Tensor lse, at::Tensor dloss,
                             long ignore_index);

void bind_aio(py::module_& m);
at::Tensor gemv_bf16(at::Tensor W, at::Tensor x);
at::Tensor spatial_bias_add(at::Tensor a, at::Tensor bias,
                            c10::optional<at::Tensor> other);
std::vector<at::Tensor> quantize_int8(at::Tensor x, long group_size);
at::Tensor dequantize_int8(at::Tensor q, at::Tensor scales, long group_size);
std::vector<at::Tensor> quantize_fp8(at::Tensor x, long group_size);
at::Tensor dequantize_fp8(at::Tensor q, at::Tensor scales, long group_size);
std::vector<at::Tensor> quantize_int4(at::Tensor x, long group);
at::Tensor dequantize_int4(at::Tensor q, at::Tensor scales, long group,
                           long numel);
std::vector<at::Tensor> quantize_fp_em(at::Tensor x, long qbits, long group);
at::Tensor dequantize_fp_em(at::Tensor q, at::Tensor scales, long qbits,
                            long group, std::vector<long> shape);

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  bind_aio(m);
  m.def("multi_tensor_adam", &multi_tensor_adam, "fused AdamW (gfx950)",
        py::arg("params"), py::arg("grads"), py::arg("exp_avgs"),
        py::arg("exp_avg_sqs"), py::arg("lr"), py::arg("beta1"),
        py::arg("beta2"), py::arg("eps"), py::arg("step"),
        py::arg("adamw_mode"), py::arg("bias_correction"),
        py::arg("weight_decay"), py::arg("out16") = std::vector<at::Tensor>(),
        py::arg("grad_scale") = 1.0);
  m.def("accum_bf16_to_f32", &accum_bf16_to_f32, py::arg("dst"),
        py::arg("src"), py::arg("scale") = 1.0);
  m.def("l2norm_sq", &l2norm_sq);
  m.def("fp8_amax", &fp8_amax);
  m.def("ragged_decode", &ragged_decode, py::arg("q"), py::arg("kpool"),
        py::arg("vpool"), py::arg("rows"), py::arg("lens"),
        py::arg("chunk") = 512);
  m.def("fp8_cast", &fp8_cast, py::arg("x"), py::arg("scale"),
        py::arg("e5m2") = false);
  m.def("fp8_cast_transpose", &fp8_cast_transpose, py::arg("x"),
        py::arg("scale"), py::arg("e5m2") = false);
  m.def("gemv_bf16", &gemv_bf16);
  m.def("spatial_bias_add", &spatial_bias_add, py::arg("a"), py::arg("bias"),
        py::arg("other") = c10::nullopt);
  m.def("quantize_int8", &quantize_int8);
  m.def("dequantize_int8", &dequantize_int8);
  m.def("quantize_fp8", &quantize_fp8);
  m.def("dequantize_fp8", &dequantize_fp8);
  m.def("quantize_int4", &quantize_int4, py::arg("x"),
        py::arg("group") = 2048);
  m.def("dequantize_int4", &dequantize_int4, py::arg("q"),
        py::arg("scales"), py::arg("group"), py::arg("numel"));
  m.def("quantize_fp_em", &quantize_fp_em, py::arg("x"), py::arg("qbits"),
        py::arg("group") = 512);
  m.def("dequantize_fp_em", &dequantize_fp_em, py::arg("q"),
        py::arg("scales"), py::arg("qbits"), py::arg("group"),
        py::arg("shape"));
  m.def("flash_attn_fwd", &flash_attn_fwd, py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("causal") = true, py::arg("scale") = 0.0,
        py::arg("kvmask") = c10::nullopt);
  m.def("flash_attn_bwd", &flash_attn_bwd, py::arg("q"), py::arg("k"),
        py::arg("v"), py::arg("dout"), py::arg("out"), py::arg("lse"),
        py::arg("causal") = true, py::arg("scale") = 0.0,
        py::arg("kvmask") = c10::nullopt);
  m.def("multi_tensor_adagrad", &multi_tensor_adagrad, py::arg("params"),
        py::arg("grads"), py::arg("sq_accums"), py::arg("lr"),
        py::arg("eps") = 1e-8, py::arg("weight_decay") = 0.0,
        py::arg("out16") = std::vector<at::Tensor>{});
  m.def("multi_tensor_lamb", &multi_tensor_lamb, py::arg("params"),
        py::arg("grads"), py::arg("updates"), py::arg("exp_avgs"),
        py::arg("exp_avg_sqs"),
        py::arg("workspace"), py::arg("lr"), py::arg("beta1") = 0.9,
        py::arg("beta2") = 0.999, py::arg("eps") = 1e-6, py::arg("step") = 1,
        py::arg("bias_correction") = 1, py::arg("weight_decay") = 0.0,
        py::arg("out16") = std::vector<at::Tensor>{},
        py::arg("grad_scale") = 1.0);
  m.def("cpu_lion_step", &cpu_lion_step, py::arg("p"), py::arg("g"),
        py::arg("m"), py::arg("out16") = c10::nullopt, py::arg("lr") = 1e-4,
        py::arg("beta1") = 0.9, py::arg("beta2") = 0.99, py::arg("wd") = 0.0,
        py::call_guard<py::gil_scoped_release>());
  m.def("cpu_adagrad_step", &cpu_adagrad_step, py::arg("p"), py::arg("g"),
        py::arg("h"), py::arg("out16") = c10::nullopt, py::arg("lr") = 1e-2,
        py::arg("eps") = 1e-8, py::arg("wd") = 0.0,
        py::call_guard<py::gil_scoped_release>());
  m.def("cpu_adam_step", &cpu_adam_step, py::arg("p"), py::arg("g"),
        py::arg("m"), py::arg("v"), py::arg("out16") = c10::nullopt,
        py::arg("lr") = 1e-3, py::arg("beta1") = 0.9, py::arg("beta2") = 0.999,
        py::arg("eps") = 1e-8, py::arg("step") = 1, py::arg("adamw") = 1,
        py::arg("bias_correction") = 1, py::arg("wd") = 0.0,
        py::arg("grad_scale") = 1.0);
  m.def("multi_tensor_lion", &multi_tensor_lion, "fused Lion (gfx950)",
        py::arg("params"), py::arg("grads"), py::arg("exp_avgs"),
        py::arg("lr"), py::arg("beta1"), py::arg("beta2"),
        py::arg("weight_decay"), py::arg("out16") = std::vector<at::Tensor>());
  m.def("rmsnorm_fwd", &rmsnorm_fwd);
  m.def("rmsnorm_bwd", &rmsnorm_bwd);
  m.def("layernorm_fwd", &layernorm_fwd);
  m.def("layernorm_bwd", &layernorm_bwd);
  m.def("rope", &rope);
  m.def("rope_inplace", &rope_inplace);
  m.def("swiglu_fwd", &swiglu_fwd);
  m.def("swiglu_bwd", &swiglu_bwd);
  m.def("cross_entropy_fwd", &cross_entropy_fwd);
  m.def("cross_entropy_bwd", &cross_entropy_bwd);
}
