// Skinny GEMV for single-token decode (MI355X, gfx950).
//
// y[B,N] = x[B,K] @ W[N,K]^T with B tiny (decode batch). hipBLASLt's tile
// GEMM shapes underutilize HBM at M=1 (~2.1 TB/s measured on 8B decode);
// this kernel streams W at full bandwidth: one wave per output row,
// bf16x8 lane loads, wave reduction. x is L2-resident (read by all rows).
#include <torch/extension.h>

#include "common.h"

// block = 256 (4 waves); wave w of block handles row (blockIdx.x*4 + w).
__global__ void gemv_bf16_kernel(const short* __restrict__ W,  // [N,K]
                                 const short* __restrict__ x,  // [B,K]
                                 short* __restrict__ y,        // [B,N]
                                 int N, int K) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row = blockIdx.x * 4 + wave;
  const int b = blockIdx.y;
  if (row >= N) return;
  const short* wr = W + (long long)row * K;
  const short* xr = x + (long long)b * K;
  float acc = 0.f;
  const int K8 = K / 8;
  for (int i = lane; i < K8; i += 64) {
    bf16x8 wv = reinterpret_cast<const bf16x8*>(wr)[i];
    bf16x8 xv = reinterpret_cast<const bf16x8*>(xr)[i];
#pragma unroll
    for (int k = 0; k < 8; ++k) acc += bf2f(wv.v[k]) * bf2f(xv.v[k]);
  }
  for (int i = K8 * 8 + lane; i < K; i += 64)
    acc += bf2f(wr[i]) * bf2f(xr[i]);
  acc = wave_reduce_sum(acc);
  if (lane == 0) y[(long long)b * N + row] = f2bf(acc);
}

// split-K variant for small N (grid would underfill 256 CUs otherwise):
// whole 256-thread block on ONE row, LDS-combined partial sums.
__global__ void gemv_bf16_splitk_kernel(const short* __restrict__ W,
                                        const short* __restrict__ x,
                                        short* __restrict__ y, int N,
                                        int K) {
  __shared__ float lds[4];
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int row = blockIdx.x;
  const int b = blockIdx.y;
  const short* wr = W + (long long)row * K;
  const short* xr = x + (long long)b * K;
  float acc = 0.f;
  const int K8 = K / 8;
  for (int i = threadIdx.x; i < K8; i += 256) {
    bf16x8 wv = reinterpret_cast<const bf16x8*>(wr)[i];
    bf16x8 xv = reinterpret_cast<const bf16x8*>(xr)[i];
#pragma unroll
    for (int k = 0; k < 8; ++k) acc += bf2f(wv.v[k]) * bf2f(xv.v[k]);
  }
  for (int i = K8 * 8 + threadIdx.x; i < K; i += 256)
    acc += bf2f(wr[i]) * bf2f(xr[i]);
  acc = wave_reduce_sum(acc);
  if (lane == 0) lds[wave] = acc;
  __syncthreads();
  if (threadIdx.x == 0)
    y[(long long)b * N + row] =
        f2bf(lds[0] + lds[1] + lds[2] + lds[3]);
}

at::Tensor gemv_bf16(at::Tensor W, at::Tensor x) {
  TORCH_CHECK(W.scalar_type() == at::kBFloat16 && W.is_contiguous());
  TORCH_CHECK(x.scalar_type() == at::kBFloat16 && x.is_contiguous());
  int N = W.size(0), K = W.size(1);
  int B = x.numel() / K;
  auto y = at::empty({B, N}, x.options());
  auto stream = c10::hip::getCurrentHIPStream();
  if (N < 4096) {
    dim3 grid(N, B);
    hipLaunchKernelGGL(gemv_bf16_splitk_kernel, grid, dim3(256), 0,
                       stream.stream(),
                       reinterpret_cast<const short*>(W.data_ptr()),
                       reinterpret_cast<const short*>(x.data_ptr()),
                       reinterpret_cast<short*>(y.data_ptr()), N, K);
  } else {
    dim3 grid((N + 3) / 4, B);
    hipLaunchKernelGGL(gemv_bf16_kernel, grid, dim3(256), 0, stream.stream(),
                       reinterpret_cast<const short*>(W.data_ptr()),
                       reinterpret_cast<const short*>(x.data_ptr()),
                       reinterpret_cast<short*>(y.data_ptr()), N, K);
  }
  HIP_CHECK_KERNEL();
  return y;
}
