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

// ---------------------------------------------------------------------------
// Generic E/M float codec (OCP MX conventions: no inf/nan encodings, all-ones
// exponent is finite, saturating). fp4 = e2m1, fp6 = e3m2, fp12 = e5m6 —
// the first two match gfx950's MX microscaling element formats.
// Parity role: reference csrc/fp_quantizer (FP_Quantize q_bits 4/6/8/12).
// Containers are byte-aligned (fp4 packs 2/byte; fp6 -> u8; fp12 -> u16).
// ---------------------------------------------------------------------------
DEV_INLINE unsigned int enc_fp_em(float v, int E, int M) {
  const int bias = (1 << (E - 1)) - 1;
  const int efield_max = (1 << E) - 1;
  const float maxval =
      (2.f - 1.f / (1 << M)) * exp2f((float)(efield_max - bias));
  unsigned int s = v < 0.f ? 1u : 0u;
  float a = fabsf(v);
  if (a != a) a = maxval;        // nan -> saturate
  if (a > maxval) a = maxval;
  if (a == 0.f) return s << (E + M);
  int e;
  float m = frexpf(a, &e);       // a = m * 2^e, m in [0.5, 1)
  int ebits = e - 1 + bias;      // exponent field if normal
  unsigned int q;
  if (ebits >= 1) {
    q = (unsigned int)rintf((2.f * m - 1.f) * (1 << M));
    if (q == (1u << M)) {        // mantissa rounding carried out
      q = 0;
      ++ebits;
    }
    if (ebits > efield_max) {    // saturate
      ebits = efield_max;
      q = (1u << M) - 1;
    }
  } else {                       // subnormal: value = q * 2^(1-bias-M)
    q = (unsigned int)rintf(a * exp2f((float)(bias - 1 + M)));
    ebits = 0;
    if (q >= (1u << M)) {        // rounded up into the first normal
      ebits = 1;
      q = 0;
    }
  }
  return (s << (E + M)) | ((unsigned int)ebits << M) | q;
}

DEV_INLINE float dec_fp_em(unsigned int bits, int E, int M) {
  const int bias = (1 << (E - 1)) - 1;
  unsigned int s = (bits >> (E + M)) & 1u;
  unsigned int ef = (bits >> M) & ((1u << E) - 1u);
  unsigned int mf = bits & ((1u << M) - 1u);
  float v;
  if (ef == 0) {
    v = (float)mf * exp2f((float)(1 - bias - M));
  } else {
    v = (1.f + (float)mf / (1 << M)) * exp2f((float)((int)ef - bias));
  }
  return s ? -v : v;
}

// group-wise: scale = group_absmax / fmt_max, stored fp32; q = enc(x/scale)
template <int E, int M>
__global__ void quantize_fp_em_kernel(const short* __restrict__ x,
                                      unsigned short* __restrict__ q,
                                      float* __restrict__ scales,
                                      long long n, int group) {
  const int bias = (1 << (E - 1)) - 1;
  const float maxval =
      (2.f - 1.f / (1 << M)) * exp2f((float)(((1 << E) - 1) - bias));
  long long g = blockIdx.x;
  const short* xg = x + g * group;
  unsigned short* qg = q + g * group;
  long long rem = n - g * group;
  int len = rem < group ? (int)rem : group;
  __shared__ float red[4];
  float amax = 0.f;
  for (int i = threadIdx.x; i < len; i += blockDim.x)
    amax = fmaxf(amax, fabsf(bf2f(xg[i])));
  amax = block_reduce_max<256>(amax, red);
  __shared__ float s_scale;
  if (threadIdx.x == 0) {
    float sc = amax > 0.f ? amax / maxval : 1.f;
    s_scale = sc;
    scales[g] = sc;
  }
  __syncthreads();
  // divide (not reciprocal-multiply): keeps bit-parity with the host codec
  for (int i = threadIdx.x; i < len; i += blockDim.x)
    qg[i] = (unsigned short)enc_fp_em(bf2f(xg[i]) / s_scale, E, M);
}

template <int E, int M>
__global__ void dequantize_fp_em_kernel(const unsigned short* __restrict__ q,
                                        const float* __restrict__ scales,
                                        short* __restrict__ x, long long n,
                                        int group) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  float sc = scales[i / group];
  x[i] = f2bf(dec_fp_em(q[i], E, M) * sc);
}

static void _launch_fp_em(int qbits, const at::Tensor& x, at::Tensor& q,
                          at::Tensor& scales, long long n, int group,
                          long long ngroups, hipStream_t stream) {
  const short* xp = reinterpret_cast<const short*>(x.data_ptr());
  unsigned short* qp = reinterpret_cast<unsigned short*>(q.data_ptr());
  float* sp = scales.data_ptr<float>();
  if (qbits == 4) {
    hipLaunchKernelGGL((quantize_fp_em_kernel<2, 1>), dim3(ngroups),
                       dim3(256), 0, stream, xp, qp, sp, n, group);
  } else if (qbits == 6) {
    hipLaunchKernelGGL((quantize_fp_em_kernel<3, 2>), dim3(ngroups),
                       dim3(256), 0, stream, xp, qp, sp, n, group);
  } else {
    hipLaunchKernelGGL((quantize_fp_em_kernel<5, 6>), dim3(ngroups),
                       dim3(256), 0, stream, xp, qp, sp, n, group);
  }
}

std::vector<at::Tensor> quantize_fp_em(at::Tensor x, long qbits, long group) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(qbits == 4 || qbits == 6 || qbits == 12);
  long long n = x.numel();
  long long ngroups = (n + group - 1) / group;
  auto q = at::empty({n}, x.options().dtype(at::kUInt16));
  auto scales = at::empty({ngroups}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  _launch_fp_em((int)qbits, x, q, scales, n, (int)group, ngroups,
                stream.stream());
  HIP_CHECK_KERNEL();
  return {q, scales};
}

at::Tensor dequantize_fp_em(at::Tensor q, at::Tensor scales, long qbits,
                            long group, std::vector<long> shape) {
  long long n = q.numel();
  auto x = at::empty({n}, q.options().dtype(at::kBFloat16));
  auto stream = c10::hip::getCurrentHIPStream();
  int block = 256;
  long long grid = (n + block - 1) / block;
  const unsigned short* qp =
      reinterpret_cast<const unsigned short*>(q.data_ptr());
  short* xp = reinterpret_cast<short*>(x.data_ptr());
  if (qbits == 4) {
    hipLaunchKernelGGL((dequantize_fp_em_kernel<2, 1>), dim3(grid),
                       dim3(block), 0, stream.stream(), qp,
                       scales.data_ptr<float>(), xp, n, (int)group);
  } else if (qbits == 6) {
    hipLaunchKernelGGL((dequantize_fp_em_kernel<3, 2>), dim3(grid),
                       dim3(block), 0, stream.stream(), qp,
                       scales.data_ptr<float>(), xp, n, (int)group);
  } else {
    hipLaunchKernelGGL((dequantize_fp_em_kernel<5, 6>), dim3(grid),
                       dim3(block), 0, stream.stream(), qp,
                       scales.data_ptr<float>(), xp, n, (int)group);
  }
  HIP_CHECK_KERNEL();
  std::vector<int64_t> sh(shape.begin(), shape.end());
  return x.reshape(sh);
}

// symmetric int4, packed two nibbles per byte, per-group fp scale
__global__ void quantize_int4_kernel(const short* __restrict__ x,
                                     unsigned char* __restrict__ q,
                                     float* __restrict__ scales,
                                     long long n, int group) {
  long long g = blockIdx.x;
  const short* xg = x + g * group;
  unsigned char* qg = q + g * (group / 2);
  long long rem = n - g * group;
  int len = rem < group ? (int)rem : group;
  __shared__ float red[4];
  float amax = 0.f;
  for (int i = threadIdx.x; i < len; i += blockDim.x)
    amax = fmaxf(amax, fabsf(bf2f(xg[i])));
  amax = block_reduce_max<256>(amax, red);
  __shared__ float s_scale;
  if (threadIdx.x == 0) {
    float sc = amax > 0.f ? amax / 7.f : 1.f;
    s_scale = sc;
    scales[g] = sc;
  }
  __syncthreads();
  float inv = 1.f / s_scale;
  for (int i = threadIdx.x; i * 2 < len; i += blockDim.x) {
    int a = (int)rintf(bf2f(xg[2 * i]) * inv);
    a = a < -7 ? -7 : (a > 7 ? 7 : a);
    int b = 0;
    if (2 * i + 1 < len) {
      b = (int)rintf(bf2f(xg[2 * i + 1]) * inv);
      b = b < -7 ? -7 : (b > 7 ? 7 : b);
    }
    qg[i] = (unsigned char)((a & 0xf) | ((b & 0xf) << 4));
  }
}

__global__ void dequantize_int4_kernel(const unsigned char* __restrict__ q,
                                       const float* __restrict__ scales,
                                       short* __restrict__ x, long long n,
                                       int group) {
  long long i = (long long)blockIdx.x * blockDim.x + threadIdx.x;
  if (2 * i >= n) return;
  float sc = scales[(2 * i) / group];
  unsigned char byte = q[i];
  int a = (int)(byte & 0xf);
  if (a > 7) a -= 16;
  int b = (int)(byte >> 4);
  if (b > 7) b -= 16;
  x[2 * i] = f2bf(a * sc);
  if (2 * i + 1 < n) x[2 * i + 1] = f2bf(b * sc);
}

std::vector<at::Tensor> quantize_int4(at::Tensor x, long group) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  TORCH_CHECK(group % 2 == 0);
  long long n = x.numel();
  long long ngroups = (n + group - 1) / group;
  auto q = at::empty({ngroups * (group / 2)}, x.options().dtype(at::kByte));
  auto scales = at::empty({ngroups}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  hipLaunchKernelGGL(quantize_int4_kernel, dim3(ngroups), dim3(256), 0,
                     stream.stream(),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     reinterpret_cast<unsigned char*>(q.data_ptr()),
                     scales.data_ptr<float>(), n, (int)group);
  HIP_CHECK_KERNEL();
  return {q, scales};
}

at::Tensor dequantize_int4(at::Tensor q, at::Tensor scales, long group,
                           long numel) {
  auto x = at::empty({numel}, q.options().dtype(at::kBFloat16));
  auto stream = c10::hip::getCurrentHIPStream();
  long long pairs = (numel + 1) / 2;
  int block = 256;
  long long grid = (pairs + block - 1) / block;
  hipLaunchKernelGGL(dequantize_int4_kernel, dim3(grid), dim3(block), 0,
                     stream.stream(),
                     reinterpret_cast<const unsigned char*>(q.data_ptr()),
                     scales.data_ptr<float>(),
                     reinterpret_cast<short*>(x.data_ptr()), numel,
                     (int)group);
  HIP_CHECK_KERNEL();
  return x;
}
