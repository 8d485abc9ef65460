// Fused softmax cross-entropy over large vocabularies (MI355X, gfx950).
//
// Role parity: the reference computes LM loss in framework Python (and tiles
// it via TiledFusedLogitsLoss, runtime/sequence_parallel/ulysses_sp.py:1065).
// MI355X-native: one workgroup per row, ONLINE max/sum in a single pass over
// V (bf16x8 vector loads), so a 128k-vocab row costs one read in fwd and one
// read+write in bwd; fp32 accumulation; optional ignore_index.
#include <torch/extension.h>

#include "common.h"

template <int BLOCK>
__global__ void ce_fwd_kernel(const short* __restrict__ logits,
                              const long* __restrict__ targets,
                              float* __restrict__ loss,
                              float* __restrict__ lse_out, int V,
                              long ignore_index) {
  __shared__ float lds[BLOCK / WAVE];
  const long long row = blockIdx.x;
  const short* xr = logits + row * (long long)V;
  long tgt = targets[row];
  if (tgt == ignore_index) {
    if (threadIdx.x == 0) {
      loss[row] = 0.f;
      lse_out[row] = 0.f;
    }
    return;
  }
  int V8 = V / 8;
  // online max + sum(exp(x - max)) in one pass
  float m = -INFINITY, s = 0.f;
  for (int i = threadIdx.x; i < V8; i += BLOCK) {
    bf16x8 v = reinterpret_cast<const bf16x8*>(xr)[i];
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      float f = bf2f(v.v[k]);
      if (f > m) {
        s *= __expf(m - f);
        m = f;
      }
      s += __expf(f - m);
    }
  }
  for (int i = V8 * 8 + threadIdx.x; i < V; i += BLOCK) {
    float f = bf2f(xr[i]);
    if (f > m) {
      s *= __expf(m - f);
      m = f;
    }
    s += __expf(f - m);
  }
  // block combine of (m, s) pairs
  float gm = block_reduce_max<BLOCK>(m, lds);
  s *= __expf(m - gm);
  float gs = block_reduce_sum<BLOCK>(s, lds);
  float lse = gm + __logf(gs);
  if (threadIdx.x == 0) {
    lse_out[row] = lse;
    loss[row] = lse - bf2f(xr[tgt]);
  }
}

// dlogits[row, c] = dloss[row] * (softmax - onehot); writes in a fresh
// buffer (or could be done in-place over logits by the caller's choice).
__global__ void ce_bwd_kernel(const short* __restrict__ logits,
                              const long* __restrict__ targets,
                              const float* __restrict__ lse,
                              const float* __restrict__ dloss,
                              short* __restrict__ dlogits, int V,
                              long ignore_index) {
  const long long row = blockIdx.y;
  const short* xr = logits + row * (long long)V;
  short* dr = dlogits + row * (long long)V;
  long tgt = targets[row];
  float dl = (tgt == ignore_index) ? 0.f : dloss[row];
  float l = lse[row];
  int i0 = blockIdx.x * blockDim.x + threadIdx.x;
  int stride = gridDim.x * blockDim.x;
  int V8 = V / 8;
  for (int i = i0; i < V8; i += stride) {
    bf16x8 v = reinterpret_cast<const bf16x8*>(xr)[i];
    bf16x8 o;
#pragma unroll
    for (int k = 0; k < 8; ++k) {
      int c = i * 8 + k;
      float p = __expf(bf2f(v.v[k]) - l);
      float g = dl * (p - (c == (int)tgt ? 1.f : 0.f));
      o.v[k] = f2bf(g);
    }
    reinterpret_cast<bf16x8*>(dr)[i] = o;
  }
  for (int i = V8 * 8 + i0; i < V; i += stride) {
    float p = __expf(bf2f(xr[i]) - l);
    dr[i] = f2bf(dl * (p - (i == (int)tgt ? 1.f : 0.f)));
  }
}

std::vector<at::Tensor> cross_entropy_fwd(at::Tensor logits,
                                          at::Tensor targets,
                                          long ignore_index) {
  TORCH_CHECK(logits.scalar_type() == at::kBFloat16 && logits.is_contiguous());
  TORCH_CHECK(targets.scalar_type() == at::kLong);
  int V = logits.size(-1);
  long long rows = logits.numel() / V;
  auto loss = at::empty({rows}, logits.options().dtype(at::kFloat));
  auto lse = at::empty({rows}, logits.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  constexpr int BLOCK = 256;
  hipLaunchKernelGGL(ce_fwd_kernel<BLOCK>, dim3(rows), dim3(BLOCK), 0,
                     stream.stream(),
                     reinterpret_cast<const short*>(logits.data_ptr()),
                     targets.data_ptr<long>(), loss.data_ptr<float>(),
                     lse.data_ptr<float>(), V, ignore_index);
  HIP_CHECK_KERNEL();
  return {loss, lse};
}

at::Tensor cross_entropy_bwd(at::Tensor logits, at::Tensor targets,
                             at::Tensor lse, at::Tensor dloss,
                             long ignore_index) {
  int V = logits.size(-1);
  long long rows = logits.numel() / V;
  auto dlogits = at::empty_like(logits);
  auto stream = c10::hip::getCurrentHIPStream();
  int bx = (V / 8 + 255) / 256;
  if (bx > 32) bx = 32;
  if (bx < 1) bx = 1;
  hipLaunchKernelGGL(ce_bwd_kernel, dim3(bx, rows), dim3(256), 0,
                     stream.stream(),
                     reinterpret_cast<const short*>(logits.data_ptr()),
                     targets.data_ptr<long>(), lse.data_ptr<float>(),
                     dloss.data_ptr<float>(),
                     reinterpret_cast<short*>(dlogits.data_ptr()), V,
                     ignore_index);
  HIP_CHECK_KERNEL();
  return dlogits;
}
