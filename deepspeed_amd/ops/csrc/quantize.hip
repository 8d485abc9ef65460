// Group-wise quantizers for MI355X (gfx950): symmetric int8/int4 and
// OCP fp8-e4m3 with per-group scales.
//
// Role parity: reference csrc/quantization/ (quantize.cu, quantize_intX.cu)
// and csrc/fp_quantizer/ (fp_quantize_impl.cu). Feeds ZeRO++-style
// quantized collectives and quantized-weight inference (linear/quantization).
// gfx950 note: fp8 is OCP e4m3fn (guide §4) — matches __hip_fp8_e4m3.
#include <torch/extension.h>
#include <hip/hip_fp8.h>

#include "common.h"

// one workgroup per group; group sizes up to 4096
template <int BLOCK>
__global__ void quantize_int8_kernel(const short* __restrict__ x,
                                     signed char* __restrict__ q,
                                     float* __restrict__ scales,
                                     int group_size) {
  __shared__ float lds[BLOCK / WAVE];
  const long long g = blockIdx.x;
  const short* xg = x + g * (long long)group_size;
  signed char* qg = q + g * (long long)group_size;
  float amax = 0.f;
  for (int i = threadIdx.x; i < group_size; i += BLOCK)
    amax = fmaxf(amax, fabsf(bf2f(xg[i])));
  amax = block_reduce_max<BLOCK>(amax, lds);
  float scale = amax > 0.f ? amax / 127.f : 1.f;
  if (threadIdx.x == 0) scales[g] = scale;
  float inv = 1.f / scale;
  for (int i = threadIdx.x; i < group_size; i += BLOCK) {
    float v = bf2f(xg[i]) * inv;
    int iv = (int)rintf(v);
    iv = iv > 127 ? 127 : (iv < -127 ? -127 : iv);
    qg[i] = (signed char)iv;
  }
}

__global__ void dequantize_int8_kernel(const signed char* __restrict__ q,
                                       const float* __restrict__ scales,
                                       short* __restrict__ x,
                                       int group_size, long long total) {
  long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < total; i += stride) {
    long long g = i / group_size;
    x[i] = f2bf((float)q[i] * scales[g]);
  }
}

template <int BLOCK>
__global__ void quantize_fp8_kernel(const short* __restrict__ x,
                                    unsigned char* __restrict__ q,
                                    float* __restrict__ scales,
                                    int group_size) {
  __shared__ float lds[BLOCK / WAVE];
  const long long g = blockIdx.x;
  const short* xg = x + g * (long long)group_size;
  unsigned char* qg = q + g * (long long)group_size;
  float amax = 0.f;
  for (int i = threadIdx.x; i < group_size; i += BLOCK)
    amax = fmaxf(amax, fabsf(bf2f(xg[i])));
  amax = block_reduce_max<BLOCK>(amax, lds);
  // e4m3fn max normal = 448
  float scale = amax > 0.f ? amax / 448.f : 1.f;
  if (threadIdx.x == 0) scales[g] = scale;
  float inv = 1.f / scale;
  for (int i = threadIdx.x; i < group_size; i += BLOCK) {
    __hip_fp8_e4m3 f8(bf2f(xg[i]) * inv);
    qg[i] = f8.__x;
  }
}

__global__ void dequantize_fp8_kernel(const unsigned char* __restrict__ q,
                                      const float* __restrict__ scales,
                                      short* __restrict__ x, int group_size,
                                      long long total) {
  long long i0 = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  long long stride = (long long)gridDim.x * blockDim.x;
  for (long long i = i0; i < total; i += stride) {
    long long g = i / group_size;
    __hip_fp8_e4m3 f8;
    f8.__x = q[i];
    x[i] = f2bf((float)f8 * scales[g]);
  }
}

std::vector<at::Tensor> quantize_int8(at::Tensor x, long group_size) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  long long total = x.numel();
  TORCH_CHECK(total % group_size == 0, "numel % group_size != 0");
  long long groups = total / group_size;
  auto q = at::empty({total}, x.options().dtype(at::kChar));
  auto scales = at::empty({groups}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  constexpr int BLOCK = 256;
  hipLaunchKernelGGL(quantize_int8_kernel<BLOCK>, dim3(groups), dim3(BLOCK),
                     0, stream.stream(),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     q.data_ptr<signed char>(), scales.data_ptr<float>(),
                     (int)group_size);
  HIP_CHECK_KERNEL();
  return {q, scales};
}

at::Tensor dequantize_int8(at::Tensor q, at::Tensor scales, long group_size) {
  long long total = q.numel();
  auto x = at::empty({total}, q.options().dtype(at::kBFloat16));
  auto stream = c10::hip::getCurrentHIPStream();
  int grid = grid_for(total, 256);
  hipLaunchKernelGGL(dequantize_int8_kernel, dim3(grid), dim3(256), 0,
                     stream.stream(), q.data_ptr<signed char>(),
                     scales.data_ptr<float>(),
                     reinterpret_cast<short*>(x.data_ptr()),
                     (int)group_size, total);
  HIP_CHECK_KERNEL();
  return x;
}

std::vector<at::Tensor> quantize_fp8(at::Tensor x, long group_size) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  long long total = x.numel();
  TORCH_CHECK(total % group_size == 0);
  long long groups = total / group_size;
  auto q = at::empty({total}, x.options().dtype(at::kByte));
  auto scales = at::empty({groups}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  constexpr int BLOCK = 256;
  hipLaunchKernelGGL(quantize_fp8_kernel<BLOCK>, dim3(groups), dim3(BLOCK),
                     0, stream.stream(),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     q.data_ptr<unsigned char>(), scales.data_ptr<float>(),
                     (int)group_size);
  HIP_CHECK_KERNEL();
  return {q, scales};
}

at::Tensor dequantize_fp8(at::Tensor q, at::Tensor scales, long group_size) {
  long long total = q.numel();
  auto x = at::empty({total}, q.options().dtype(at::kBFloat16));
  auto stream = c10::hip::getCurrentHIPStream();
  int grid = grid_for(total, 256);
  hipLaunchKernelGGL(dequantize_fp8_kernel, dim3(grid), dim3(256), 0,
                     stream.stream(), q.data_ptr<unsigned char>(),
                     scales.data_ptr<float>(),
                     reinterpret_cast<short*>(x.data_ptr()),
                     (int)group_size, total);
  HIP_CHECK_KERNEL();
  return x;
}
