// Host-side fused AdamW for ZeRO-Offload (EPYC host of the MI355X node).
//
// Role parity: reference csrc/adam/cpu_adam_impl.cpp (AVX512 Step_1/4/8).
// MI355X-native: plain vectorizer-friendly fp32 loop under OpenMP — clang
// auto-vectorizes to AVX2/AVX512 at -O3; DRAM-bandwidth-bound either way.
// Optional fused bf16 writeback fills the pinned staging buffer that the
// caller hipMemcpyAsync's to the device (one H2D of 2 bytes/param).
#include <torch/extension.h>

#include <cmath>
#include <cstdint>

static inline uint16_t f32_to_bf16(float f) {
  union {
    float f;
    uint32_t u;
  } cvt;
  cvt.f = f;
  uint32_t lsb = (cvt.u >> 16) & 1;
  return (uint16_t)((cvt.u + 0x7fff + lsb) >> 16);
}

void cpu_adam_step(at::Tensor p, at::Tensor g, at::Tensor m, at::Tensor v,
                   c10::optional<at::Tensor> out16, double lr_, double beta1_,
                   double beta2_, double eps_, long step, long adamw,
                   long bias_correction, double wd_, double grad_scale_) {
  TORCH_CHECK(p.device().is_cpu() && p.scalar_type() == at::kFloat);
  TORCH_CHECK(p.is_contiguous() && g.is_contiguous());
  const float lr = (float)lr_, beta1 = (float)beta1_, beta2 = (float)beta2_;
  const float eps = (float)eps_, wd = (float)wd_;
  const float gscale = (float)grad_scale_;
  float bc1 = 1.f, bc2 = 1.f;
  if (bias_correction) {
    bc1 = 1.f - powf(beta1, (float)step);
    bc2 = 1.f - powf(beta2, (float)step);
  }
  const float step_size = lr / bc1;
  const float bc2_sqrt = sqrtf(bc2);
  const int64_t n = p.numel();
  float* pp = p.data_ptr<float>();
  const float* gp = g.data_ptr<float>();
  float* mp = m.data_ptr<float>();
  float* vp = v.data_ptr<float>();
  uint16_t* op = nullptr;
  if (out16.has_value()) {
    TORCH_CHECK(out16->scalar_type() == at::kBFloat16 &&
                out16->device().is_cpu() && out16->numel() == n);
    op = reinterpret_cast<uint16_t*>(out16->data_ptr());
  }
  const float decay = 1.f - lr * wd;
#pragma omp parallel for schedule(static)
  for (int64_t i = 0; i < n; ++i) {
    float gk = gp[i] * gscale;
    float pk = pp[i];
    if (adamw) {
      pk *= decay;
    } else if (wd != 0.f) {
      gk += wd * pk;
    }
    float mk = mp[i] * beta1 + gk * (1.f - beta1);
    float vk = vp[i] * beta2 + gk * gk * (1.f - beta2);
    float denom = sqrtf(vk) / bc2_sqrt + eps;
    pk -= step_size * mk / denom;
    mp[i] = mk;
    vp[i] = vk;
    pp[i] = pk;
    if (op) op[i] = f32_to_bf16(pk);
  }
}

// Host Lion for ZeRO-Offload (ref csrc/lion/cpu_lion_impl.cpp).
void cpu_lion_step(at::Tensor p, at::Tensor g, at::Tensor m,
                   c10::optional<at::Tensor> out16, double lr_, double beta1_,
                   double beta2_, double wd_) {
  TORCH_CHECK(p.device().is_cpu() && p.scalar_type() == at::kFloat);
  const float lr = (float)lr_, beta1 = (float)beta1_, beta2 = (float)beta2_;
  const float wd = (float)wd_;
  const int64_t n = p.numel();
  float* pp = p.data_ptr<float>();
  const float* gp = g.data_ptr<float>();
  float* mp = m.data_ptr<float>();
  uint16_t* op = nullptr;
  if (out16.has_value())
    op = reinterpret_cast<uint16_t*>(out16->data_ptr());
#pragma omp parallel for schedule(static)
  for (int64_t i = 0; i < n; ++i) {
    float gk = gp[i], pk = pp[i] * (1.f - lr * wd), mk = mp[i];
    float u = mk * beta1 + gk * (1.f - beta1);
    pk -= lr * (u > 0.f ? 1.f : (u < 0.f ? -1.f : 0.f));
    mp[i] = mk * beta2 + gk * (1.f - beta2);
    pp[i] = pk;
    if (op) op[i] = f32_to_bf16(pk);
  }
}

// Host Adagrad for ZeRO-Offload (ref csrc/adagrad/cpu_adagrad.cpp).
void cpu_adagrad_step(at::Tensor p, at::Tensor g, at::Tensor h,
                      c10::optional<at::Tensor> out16, double lr_,
                      double eps_, double wd_) {
  TORCH_CHECK(p.device().is_cpu() && p.scalar_type() == at::kFloat);
  const float lr = (float)lr_, eps = (float)eps_, wd = (float)wd_;
  const int64_t n = p.numel();
  float* pp = p.data_ptr<float>();
  const float* gp = g.data_ptr<float>();
  float* hp = h.data_ptr<float>();
  uint16_t* op = nullptr;
  if (out16.has_value())
    op = reinterpret_cast<uint16_t*>(out16->data_ptr());
#pragma omp parallel for schedule(static)
  for (int64_t i = 0; i < n; ++i) {
    float gk = gp[i], pk = pp[i];
    if (wd != 0.f) gk += wd * pk;
    float hk = hp[i] + gk * gk;
    pk -= lr * gk / (sqrtf(hk) + eps);
    hp[i] = hk;
    pp[i] = pk;
    if (op) op[i] = f32_to_bf16(pk);
  }
}
