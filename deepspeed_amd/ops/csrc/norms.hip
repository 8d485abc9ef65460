// Fused RMSNorm / LayerNorm fwd+bwd for MI355X (gfx950).
//
// Role parity: reference csrc/transformer/inference/csrc/rms_norm.cu and
// csrc/transformer/normalize_kernels.cu. MI355X-native: one workgroup per
// row, bf16x8 vector loads (guide G13), wave64 shuffle + LDS reductions,
// fp32 math throughout; dgamma via a separate column-sum kernel with
// lane-contiguous (coalesced) column access.
#include <torch/extension.h>

#include "common.h"

// ---------------------------------------------------------------- RMSNorm
template <int BLOCK>
__global__ void rmsnorm_fwd_kernel(const short* __restrict__ x,
                                   const short* __restrict__ w,
                                   short* __restrict__ y,
                                   float* __restrict__ rstd_out, int H,
                                   float eps) {
  __shared__ float lds[BLOCK / WAVE];
  const long long row = blockIdx.x;
  const short* xr = x + row * (long long)H;
  short* yr = y + row * (long long)H;
  int H8 = H / 8;
  float ss = 0.f;
  for (int i = threadIdx.x; i < H8; i += BLOCK) {
    bf16x8 v = reinterpret_cast<const bf16x8*>(xr)[i];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f = bf2f(v.v[k]);
      ss += f * f;
    }
  }
  for (int i = H8 * 8 + threadIdx.x; i < H; i += BLOCK) {
    float f = bf2f(xr[i]);
    ss += f * f;
  }
  ss = block_reduce_sum<BLOCK>(ss, lds);
  float rstd = rsqrtf(ss / H + eps);
  if (threadIdx.x == 0 && rstd_out) rstd_out[row] = rstd;
  for (int i = threadIdx.x; i < H8; i += BLOCK) {
    bf16x8 v = reinterpret_cast<const bf16x8*>(xr)[i];
    bf16x8 wv = reinterpret_cast<const bf16x8*>(w)[i];
    bf16x8 o;
#pragma unroll
    for (int k = 0; k < 8; ++k)
      o.v[k] = f2bf(bf2f(v.v[k]) * rstd * bf2f(wv.v[k]));
    reinterpret_cast<bf16x8*>(yr)[i] = o;
  }
  for (int i = H8 * 8 + threadIdx.x; i < H; i += BLOCK)
    yr[i] = f2bf(bf2f(xr[i]) * rstd * bf2f(w[i]));
}

// dx = rstd * (dy*w - xhat * mean(dy*w*xhat))   where xhat = x*rstd
template <int BLOCK>
__global__ void rmsnorm_bwd_dx_kernel(const short* __restrict__ dy,
                                      const short* __restrict__ x,
                                      const short* __restrict__ w,
                                      const float* __restrict__ rstd,
                                      short* __restrict__ dx, int H) {
  __shared__ float lds[BLOCK / WAVE];
  const long long row = blockIdx.x;
  const short* dyr = dy + row * (long long)H;
  const short* xr = x + row * (long long)H;
  short* dxr = dx + row * (long long)H;
  float rs = rstd[row];
  int H8 = H / 8;
  float dot = 0.f;
  for (int i = threadIdx.x; i < H8; i += BLOCK) {
    bf16x8 dv = reinterpret_cast<const bf16x8*>(dyr)[i];
    bf16x8 xv = reinterpret_cast<const bf16x8*>(xr)[i];
    bf16x8 wv = reinterpret_cast<const bf16x8*>(w)[i];
#pragma unroll
    for (int k = 0; k < 8; ++k)
      dot += bf2f(dv.v[k]) * bf2f(wv.v[k]) * bf2f(xv.v[k]);
  }
  for (int i = H8 * 8 + threadIdx.x; i < H; i += BLOCK)
    dot += bf2f(dyr[i]) * bf2f(w[i]) * bf2f(xr[i]);
  dot = block_reduce_sum<BLOCK>(dot, lds);
  float c = dot * rs * rs / H;  // mean(dy*w*x) * rstd^2
  for (int i = threadIdx.x; i < H8; i += BLOCK) {
    bf16x8 dv = reinterpret_cast<const bf16x8*>(dyr)[i];
    bf16x8 xv = reinterpret_cast<const bf16x8*>(xr)[i];
    bf16x8 wv = reinterpret_cast<const bf16x8*>(w)[i];
    bf16x8 o;
#pragma unroll
    for (int k = 0; k < 8; ++k)
      o.v[k] = f2bf(rs * (bf2f(dv.v[k]) * bf2f(wv.v[k]) -
                          bf2f(xv.v[k]) * c));
    reinterpret_cast<bf16x8*>(dxr)[i] = o;
  }
  for (int i = H8 * 8 + threadIdx.x; i < H; i += BLOCK)
    dxr[i] = f2bf(rs * (bf2f(dyr[i]) * bf2f(w[i]) - bf2f(xr[i]) * c));
}

// dw[c] = sum_r dy[r,c] * x[r,c] * rstd[r]; column-parallel, coalesced:
// consecutive lanes take consecutive columns.
__global__ void rmsnorm_bwd_dw_kernel(const short* __restrict__ dy,
                                      const short* __restrict__ x,
                                      const float* __restrict__ rstd,
                                      float* __restrict__ dw, long long rows,
                                      int H) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= H) return;
  float acc = 0.f;
  for (long long r = blockIdx.y; r < rows; r += gridDim.y) {
    long long idx = r * H + c;
    acc += bf2f(dy[idx]) * bf2f(x[idx]) * rstd[r];
  }
  if (gridDim.y == 1)
    dw[c] = acc;
  else
    atomicAdd(dw + c, acc);
}

std::vector<at::Tensor> rmsnorm_fwd(at::Tensor x, at::Tensor w, double eps) {
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  int H = x.size(-1);
  long long rows = x.numel() / H;
  auto y = at::empty_like(x);
  auto rstd = at::empty({rows}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  constexpr int BLOCK = 256;
  hipLaunchKernelGGL(rmsnorm_fwd_kernel<BLOCK>, dim3(rows), dim3(BLOCK), 0,
                     stream.stream(),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     reinterpret_cast<const short*>(w.data_ptr()),
                     reinterpret_cast<short*>(y.data_ptr()),
                     rstd.data_ptr<float>(), H, (float)eps);
  HIP_CHECK_KERNEL();
  return {y, rstd};
}

std::vector<at::Tensor> rmsnorm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                    at::Tensor rstd) {
  int H = x.size(-1);
  long long rows = x.numel() / H;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({H}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  constexpr int BLOCK = 256;
  hipLaunchKernelGGL(rmsnorm_bwd_dx_kernel<BLOCK>, dim3(rows), dim3(BLOCK), 0,
                     stream.stream(),
                     reinterpret_cast<const short*>(dy.data_ptr()),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     reinterpret_cast<const short*>(w.data_ptr()),
                     rstd.data_ptr<float>(),
                     reinterpret_cast<short*>(dx.data_ptr()), H);
  HIP_CHECK_KERNEL();
  int gy = rows > 4096 ? 64 : (rows > 512 ? 16 : 1);
  hipLaunchKernelGGL(rmsnorm_bwd_dw_kernel, dim3((H + 255) / 256, gy),
                     dim3(256), 0, stream.stream(),
                     reinterpret_cast<const short*>(dy.data_ptr()),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     rstd.data_ptr<float>(), dw.data_ptr<float>(), rows, H);
  HIP_CHECK_KERNEL();
  return {dx, dw};
}

// --------------------------------------------------------------- LayerNorm
template <int BLOCK>
__global__ void layernorm_fwd_kernel(const short* __restrict__ x,
                                     const short* __restrict__ w,
                                     const short* __restrict__ b,
                                     short* __restrict__ y,
                                     float* __restrict__ mean_out,
                                     float* __restrict__ rstd_out, int H,
                                     float eps) {
  __shared__ float lds[BLOCK / WAVE];
  const long long row = blockIdx.x;
  const short* xr = x + row * (long long)H;
  short* yr = y + row * (long long)H;
  float s = 0.f, ss = 0.f;
  for (int i = threadIdx.x; i < H; i += BLOCK) {
    float f = bf2f(xr[i]);
    s += f;
    ss += f * f;
  }
  s = block_reduce_sum<BLOCK>(s, lds);
  ss = block_reduce_sum<BLOCK>(ss, lds);
  float mean = s / H;
  float var = ss / H - mean * mean;
  float rstd = rsqrtf(var + eps);
  if (threadIdx.x == 0) {
    if (mean_out) mean_out[row] = mean;
    if (rstd_out) rstd_out[row] = rstd;
  }
  for (int i = threadIdx.x; i < H; i += BLOCK) {
    float xhat = (bf2f(xr[i]) - mean) * rstd;
    float o = xhat * bf2f(w[i]) + (b ? bf2f(b[i]) : 0.f);
    yr[i] = f2bf(o);
  }
}

template <int BLOCK>
__global__ void layernorm_bwd_dx_kernel(
    const short* __restrict__ dy, const short* __restrict__ x,
    const short* __restrict__ w, const float* __restrict__ mean,
    const float* __restrict__ rstd, short* __restrict__ dx, int H) {
  __shared__ float lds[BLOCK / WAVE];
  const long long row = blockIdx.x;
  const short* dyr = dy + row * (long long)H;
  const short* xr = x + row * (long long)H;
  short* dxr = dx + row * (long long)H;
  float mu = mean[row], rs = rstd[row];
  float c1 = 0.f, c2 = 0.f;
  for (int i = threadIdx.x; i < H; i += BLOCK) {
    float g = bf2f(dyr[i]) * bf2f(w[i]);
    float xhat = (bf2f(xr[i]) - mu) * rs;
    c1 += g;
    c2 += g * xhat;
  }
  c1 = block_reduce_sum<BLOCK>(c1, lds) / H;
  c2 = block_reduce_sum<BLOCK>(c2, lds) / H;
  for (int i = threadIdx.x; i < H; i += BLOCK) {
    float g = bf2f(dyr[i]) * bf2f(w[i]);
    float xhat = (bf2f(xr[i]) - mu) * rs;
    dxr[i] = f2bf((g - c1 - xhat * c2) * rs);
  }
}

__global__ void layernorm_bwd_dwdb_kernel(const short* __restrict__ dy,
                                          const short* __restrict__ x,
                                          const float* __restrict__ mean,
                                          const float* __restrict__ rstd,
                                          float* __restrict__ dw,
                                          float* __restrict__ db,
                                          long long rows, int H) {
  int c = blockIdx.x * blockDim.x + threadIdx.x;
  if (c >= H) return;
  float accw = 0.f, accb = 0.f;
  for (long long r = blockIdx.y; r < rows; r += gridDim.y) {
    long long idx = r * H + c;
    float d = bf2f(dy[idx]);
    accw += d * (bf2f(x[idx]) - mean[r]) * rstd[r];
    accb += d;
  }
  if (gridDim.y == 1) {
    dw[c] = accw;
    db[c] = accb;
  } else {
    atomicAdd(dw + c, accw);
    atomicAdd(db + c, accb);
  }
}

std::vector<at::Tensor> layernorm_fwd(at::Tensor x, at::Tensor w,
                                      c10::optional<at::Tensor> b,
                                      double eps) {
  int H = x.size(-1);
  long long rows = x.numel() / H;
  auto y = at::empty_like(x);
  auto mean = at::empty({rows}, x.options().dtype(at::kFloat));
  auto rstd = at::empty({rows}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  constexpr int BLOCK = 256;
  hipLaunchKernelGGL(
      layernorm_fwd_kernel<BLOCK>, dim3(rows), dim3(BLOCK), 0,
      stream.stream(), reinterpret_cast<const short*>(x.data_ptr()),
      reinterpret_cast<const short*>(w.data_ptr()),
      b.has_value() ? reinterpret_cast<const short*>(b->data_ptr()) : nullptr,
      reinterpret_cast<short*>(y.data_ptr()), mean.data_ptr<float>(),
      rstd.data_ptr<float>(), H, (float)eps);
  HIP_CHECK_KERNEL();
  return {y, mean, rstd};
}

std::vector<at::Tensor> layernorm_bwd(at::Tensor dy, at::Tensor x,
                                      at::Tensor w, at::Tensor mean,
                                      at::Tensor rstd) {
  int H = x.size(-1);
  long long rows = x.numel() / H;
  auto dx = at::empty_like(x);
  auto dw = at::zeros({H}, x.options().dtype(at::kFloat));
  auto db = at::zeros({H}, x.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  constexpr int BLOCK = 256;
  hipLaunchKernelGGL(layernorm_bwd_dx_kernel<BLOCK>, dim3(rows), dim3(BLOCK),
                     0, stream.stream(),
                     reinterpret_cast<const short*>(dy.data_ptr()),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     reinterpret_cast<const short*>(w.data_ptr()),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     reinterpret_cast<short*>(dx.data_ptr()), H);
  HIP_CHECK_KERNEL();
  int gy = rows > 4096 ? 64 : (rows > 512 ? 16 : 1);
  hipLaunchKernelGGL(layernorm_bwd_dwdb_kernel, dim3((H + 255) / 256, gy),
                     dim3(256), 0, stream.stream(),
                     reinterpret_cast<const short*>(dy.data_ptr()),
                     reinterpret_cast<const short*>(x.data_ptr()),
                     mean.data_ptr<float>(), rstd.data_ptr<float>(),
                     dw.data_ptr<float>(), db.data_ptr<float>(), rows, H);
  HIP_CHECK_KERNEL();
  return {dx, dw, db};
}
