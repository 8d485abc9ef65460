// Common device helpers for deepspeed_amd CDNA4 (gfx950) kernels.
// Hand-written HIP for MI355X: wave64, 256-CU grid sizing, vectorized IO.
#pragma once

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hip/hip_fp16.h>
#include <c10/hip/HIPStream.h>

#define WAVE 64
#define DEV_INLINE __device__ __forceinline__

// MI355X: 256 CUs; memory-bound kernels target ~8 blocks/CU then grid-stride.
constexpr int kMaxBlocks = 2048;

static inline int grid_for(long long total_threads, int block) {
  long long b = (total_threads + block - 1) / block;
  return (int)(b < kMaxBlocks ? (b > 0 ? b : 1) : kMaxBlocks);
}

using bf16 = __hip_bfloat16;

// 16-byte vector of 8 bf16 (guide G13: always vectorize bf16 loads).
struct alignas(16) bf16x8 {
  short v[8];
};
struct alignas(16) f32x4 {
  float v[4];
};

DEV_INLINE float bf2f(short b) {
  union {
    float f;
    unsigned int u;
  } cvt;
  cvt.u = ((unsigned int)(unsigned short)b) << 16;
  return cvt.f;
}

DEV_INLINE short f2bf(float f) {
#if defined(__gfx950__)
  // native RNE convert: v_cvt_pk_bf16_f32 (one VALU op vs 4 integer ops)
  __bf16 b = (__bf16)f;
  return __builtin_bit_cast(short, b);
#else
  union {
    float f;
    unsigned int u;
  } cvt;
  cvt.f = f;
  // round-to-nearest-even
  unsigned int lsb = (cvt.u >> 16) & 1;
  cvt.u += 0x7fff + lsb;
  return (short)(cvt.u >> 16);
#endif
}

// wave-level reduction over 64 lanes
// 16-lane (DPP row) reductions: row_ror rotates within each group of 16
// lanes on the VALU — no LDS-pipe traffic (vs __shfl_xor -> ds_swizzle).
// After the 4 steps every lane of the group holds the full reduction.
DEV_INLINE float dpp_ror16(float x, const int ctrl) {
  int i = __builtin_bit_cast(int, x);
  int r;
  switch (ctrl) {  // update_dpp needs literal ctrl
    case 0x128: r = __builtin_amdgcn_update_dpp(0, i, 0x128, 0xf, 0xf, true);
                break;
    case 0x124: r = __builtin_amdgcn_update_dpp(0, i, 0x124, 0xf, 0xf, true);
                break;
    case 0x122: r = __builtin_amdgcn_update_dpp(0, i, 0x122, 0xf, 0xf, true);
                break;
    default:    r = __builtin_amdgcn_update_dpp(0, i, 0x121, 0xf, 0xf, true);
                break;
  }
  return __builtin_bit_cast(float, r);
}

DEV_INLINE float row16_reduce_sum(float x) {
  x += dpp_ror16(x, 0x128);
  x += dpp_ror16(x, 0x124);
  x += dpp_ror16(x, 0x122);
  x += dpp_ror16(x, 0x121);
  return x;
}

DEV_INLINE float row16_reduce_max(float x) {
  x = fmaxf(x, dpp_ror16(x, 0x128));
  x = fmaxf(x, dpp_ror16(x, 0x124));
  x = fmaxf(x, dpp_ror16(x, 0x122));
  x = fmaxf(x, dpp_ror16(x, 0x121));
  return x;
}

DEV_INLINE float wave_reduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_down(x, off, WAVE);
  return x;
}

DEV_INLINE float wave_reduce_max(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1)
    x = fmaxf(x, __shfl_down(x, off, WAVE));
  return x;
}

// block-level reduction (block must be multiple of 64, <= 1024)
template <int BLOCK>
DEV_INLINE float block_reduce_sum(float x, float* lds /* BLOCK/64 floats */) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  x = wave_reduce_sum(x);
  if (lane == 0) lds[wid] = x;
  __syncthreads();
  constexpr int NW = BLOCK / WAVE;
  x = (threadIdx.x < NW) ? lds[threadIdx.x] : 0.f;
  if (wid == 0) {
#pragma unroll
    for (int off = NW / 2; off > 0; off >>= 1) x += __shfl_down(x, off, WAVE);
    if (lane == 0) lds[0] = x;
  }
  __syncthreads();
  float r = lds[0];
  __syncthreads();
  return r;
}

template <int BLOCK>
DEV_INLINE float block_reduce_max(float x, float* lds) {
  int lane = threadIdx.x & (WAVE - 1);
  int wid = threadIdx.x / WAVE;
  x = wave_reduce_max(x);
  if (lane == 0) lds[wid] = x;
  __syncthreads();
  constexpr int NW = BLOCK / WAVE;
  x = (threadIdx.x < NW) ? lds[threadIdx.x] : -INFINITY;
  if (wid == 0) {
#pragma unroll
    for (int off = NW / 2; off > 0; off >>= 1)
      x = fmaxf(x, __shfl_down(x, off, WAVE));
    if (lane == 0) lds[0] = x;
  }
  __syncthreads();
  float r = lds[0];
  __syncthreads();
  return r;
}

#define HIP_CHECK_KERNEL()                                         \
  do {                                                             \
    hipError_t e = hipGetLastError();                              \
    if (e != hipSuccess) {                                         \
      TORCH_CHECK(false, "HIP kernel launch failed: ",             \
                  hipGetErrorString(e));                           \
    }                                                              \
  } while (0)
