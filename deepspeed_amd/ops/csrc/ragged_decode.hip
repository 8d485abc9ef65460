// Ragged-batch decode attention for MI355X (gfx950).
//
// Role parity: reference inference/v2 ragged kernels
// (deepspeed/inference/v2/kernels/ragged_ops/, blocked KV attention,
// ragged/kv_cache.py). MI355X-native design: decode attention is
// HBM-bound (read each active sequence's KV once), so the kernel reads
// STRAIGHT from the persistent slot pool [B, Smax, Hk, D] using per-slot
// lengths — no per-token gather copies, no [n,1,1,len] masks, no padding
// (the old masked-SDPA path copied K/V slices every token and ran at
// ~2.1 TB/s effective vs the 6.3 TB/s roofline).
//
// Shape: one workgroup per (row, kv-head, kv-chunk); 4 waves split the
// chunk. Per position: one coalesced 64-lane read of the K row (lane
// owns D/64 dims), G = Hq/Hk query dots via butterfly allreduce, online
// softmax per (wave, q-head), P*V accumulated into lane-local dims.
// Chunks write fp32 partials (out, m, l); a tiny combine kernel merges
// them (flash-decoding split-KV scheme) so short batches still fill 256
// CUs.
#include <torch/extension.h>

#include "common.h"

namespace ragged {

DEV_INLINE float allreduce_sum(float x) {
#pragma unroll
  for (int off = 32; off > 0; off >>= 1) x += __shfl_xor(x, off, 64);
  return x;
}

constexpr int MAXG = 8;  // max q-heads per kv head handled per block

// q [n, Hq, D] bf16 (pre-flattened), kpool/vpool [B, Smax, Hk, D] bf16,
// rows [n] i64, lens [n] i64 (lengths AFTER appending this token),
// partial_out [n, Hq, NS, D] f32, partial_ml [n, Hq, NS, 2] f32
template <int D, int G>
__launch_bounds__(256, 8)
__global__ void ragged_decode_kernel(
    const short* __restrict__ q, const short* __restrict__ kpool,
    const short* __restrict__ vpool, const int64_t* __restrict__ rows,
    const int64_t* __restrict__ lens, float* __restrict__ partial_out,
    float* __restrict__ partial_ml, int Hq, int Hk, int Smax, int NS,
    int chunk, float scale) {
  constexpr int DL = D / 64;  // dims per lane
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int r = blockIdx.x;       // active-row index
  const int hk = blockIdx.y;      // kv head
  const int cidx = blockIdx.z;    // kv chunk
  const long long slot = rows[r];
  const int len = (int)lens[r];
  const int c0 = cidx * chunk;
  if (c0 >= len && cidx > 0) {
    // no work: mark partials empty
    if (threadIdx.x < G * 2) {
      int g = threadIdx.x / 2;
      int hq = hk * G + g;
      partial_ml[(((long long)r * Hq + hq) * NS + cidx) * 2 +
                 (threadIdx.x & 1)] = (threadIdx.x & 1) ? 0.f : -INFINITY;
    }
    return;
  }
  const int c1 = min(len, c0 + chunk);
  // wave's sub-range
  const int per_wave = (c1 - c0 + 3) / 4;
  const int w0 = c0 + wave * per_wave;
  const int w1 = min(c1, w0 + per_wave);

  const long long kvrs = (long long)Hk * D;
  const short* kbase = kpool + (slot * Smax) * kvrs + hk * D + lane * DL;
  const short* vbase = vpool + (slot * Smax) * kvrs + hk * D + lane * DL;

  // q fragments: G heads, DL dims each, pre-scaled
  float qf[G][DL];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    const short* qp = q + ((long long)r * Hq + hk * G + g) * D + lane * DL;
#pragma unroll
    for (int j = 0; j < DL; ++j) qf[g][j] = bf2f(qp[j]) * scale;
  }

  float m[G], l[G], acc[G][DL];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    m[g] = -INFINITY;
    l[g] = 0.f;
#pragma unroll
    for (int j = 0; j < DL; ++j) acc[g][j] = 0.f;
  }

  for (int p = w0; p < w1; ++p) {
    float kf[DL], vf[DL];
    const short* kr = kbase + (long long)p * kvrs;
    const short* vr = vbase + (long long)p * kvrs;
    if (DL == 2) {
      unsigned int ku = *reinterpret_cast<const unsigned int*>(kr);
      unsigned int vu = *reinterpret_cast<const unsigned int*>(vr);
      kf[0] = bf2f((short)(ku & 0xffff));
      kf[1] = bf2f((short)(ku >> 16));
      vf[0] = bf2f((short)(vu & 0xffff));
      vf[1] = bf2f((short)(vu >> 16));
    } else {
#pragma unroll
      for (int j = 0; j < DL; ++j) {
        kf[j] = bf2f(kr[j]);
        vf[j] = bf2f(vr[j]);
      }
    }
#pragma unroll
    for (int g = 0; g < G; ++g) {
      float d = 0.f;
#pragma unroll
      for (int j = 0; j < DL; ++j) d += qf[g][j] * kf[j];
      float s = allreduce_sum(d);
      if (s > m[g]) {
        float c = (m[g] == -INFINITY) ? 0.f : __expf(m[g] - s);
        l[g] *= c;
#pragma unroll
        for (int j = 0; j < DL; ++j) acc[g][j] *= c;
        m[g] = s;
      }
      float pv = __expf(s - m[g]);
      l[g] += pv;
#pragma unroll
      for (int j = 0; j < DL; ++j) acc[g][j] += pv * vf[j];
    }
  }

  // combine the 4 waves in LDS: per (g): m, l, acc[D]
  __shared__ float lm[4][MAXG], ll[4][MAXG];
  __shared__ float lacc[4][MAXG][D];
#pragma unroll
  for (int g = 0; g < G; ++g) {
    if (lane == 0) {
      lm[wave][g] = m[g];
      ll[wave][g] = l[g];
    }
#pragma unroll
    for (int j = 0; j < DL; ++j) lacc[wave][g][lane * DL + j] = acc[g][j];
  }
  __syncthreads();
  if (wave == 0) {
#pragma unroll
    for (int g = 0; g < G; ++g) {
      float mw = fmaxf(fmaxf(lm[0][g], lm[1][g]),
                       fmaxf(lm[2][g], lm[3][g]));
      float lw = 0.f;
      float o[DL];
#pragma unroll
      for (int j = 0; j < DL; ++j) o[j] = 0.f;
      for (int wv = 0; wv < 4; ++wv) {
        float mi = lm[wv][g];
        if (mi == -INFINITY) continue;
        float c = __expf(mi - mw);
        lw += ll[wv][g] * c;
#pragma unroll
        for (int j = 0; j < DL; ++j)
          o[j] += lacc[wv][g][lane * DL + j] * c;
      }
      int hq = hk * G + g;
      long long ob = (((long long)r * Hq + hq) * NS + cidx) * D + lane * DL;
#pragma unroll
      for (int j = 0; j < DL; ++j) partial_out[ob + j] = o[j];
      if (lane == 0) {
        long long mb = (((long long)r * Hq + hq) * NS + cidx) * 2;
        partial_ml[mb] = mw;
        partial_ml[mb + 1] = lw;
      }
    }
  }
}

// merge NS chunk partials -> out [n, Hq, D] bf16; one wave per (r, hq)
template <int D>
__global__ void ragged_combine_kernel(const float* __restrict__ partial_out,
                                      const float* __restrict__ partial_ml,
                                      short* __restrict__ out, long long nhq,
                                      int NS) {
  constexpr int DL = D / 64;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const long long rh = (long long)blockIdx.x * 4 + wave;  // over n*Hq
  if (rh >= nhq) return;
  float mstar = -INFINITY;
  for (int c = 0; c < NS; ++c)
    mstar = fmaxf(mstar, partial_ml[(rh * NS + c) * 2]);
  float lsum = 0.f;
  float o[DL];
#pragma unroll
  for (int j = 0; j < DL; ++j) o[j] = 0.f;
  for (int c = 0; c < NS; ++c) {
    float mi = partial_ml[(rh * NS + c) * 2];
    if (mi == -INFINITY) continue;
    float li = partial_ml[(rh * NS + c) * 2 + 1];
    float sc = __expf(mi - mstar);
    lsum += li * sc;
    const float* po = partial_out + (rh * NS + c) * D + lane * DL;
#pragma unroll
    for (int j = 0; j < DL; ++j) o[j] += po[j] * sc;
  }
  float inv = lsum > 0.f ? 1.f / lsum : 0.f;
#pragma unroll
  for (int j = 0; j < DL; ++j)
    out[rh * D + lane * DL + j] = f2bf(o[j] * inv);
}

}  // namespace ragged

at::Tensor ragged_decode(at::Tensor q, at::Tensor kpool, at::Tensor vpool,
                         at::Tensor rows, at::Tensor lens, long chunk) {
  TORCH_CHECK(q.scalar_type() == at::kBFloat16 && q.is_contiguous());
  TORCH_CHECK(kpool.is_contiguous() && vpool.is_contiguous());
  TORCH_CHECK(rows.scalar_type() == at::kLong &&
              lens.scalar_type() == at::kLong);
  int n = q.size(0);
  int Hq = q.size(-2), D = q.size(-1);
  int Smax = kpool.size(1), Hk = kpool.size(2);
  int G = Hq / Hk;
  TORCH_CHECK(Hq % Hk == 0 && G <= ragged::MAXG, "GQA group too large");
  TORCH_CHECK(D == 128 || D == 64, "head_dim must be 64 or 128");
  long maxlen = lens.max().item<long>();
  int NS = (int)std::min<long>((maxlen + chunk - 1) / chunk, 64);
  int eff_chunk = (int)((maxlen + NS - 1) / NS);
  eff_chunk = (eff_chunk + 3) & ~3;
  auto opts = q.options().dtype(at::kFloat);
  auto pout = at::empty({n, Hq, NS, D}, opts);
  auto pml = at::empty({n, Hq, NS, 2}, opts);
  auto out = at::empty({n, 1, Hq, D}, q.options());
  float scale = 1.0f / std::sqrt((float)D);
  auto stream = c10::hip::getCurrentHIPStream();
  dim3 grid(n, Hk, NS);

#define LAUNCH(DD, GG)                                                       \
  hipLaunchKernelGGL((ragged::ragged_decode_kernel<DD, GG>), grid,           \
                     dim3(256), 0, stream.stream(),                          \
                     reinterpret_cast<const short*>(q.data_ptr()),           \
                     reinterpret_cast<const short*>(kpool.data_ptr()),       \
                     reinterpret_cast<const short*>(vpool.data_ptr()),       \
                     rows.data_ptr<int64_t>(), lens.data_ptr<int64_t>(), \
                     pout.data_ptr<float>(), pml.data_ptr<float>(), Hq, Hk,  \
                     Smax, NS, eff_chunk, scale)

  bool done = false;
  if (D == 128) {
    switch (G) {
      case 1: LAUNCH(128, 1); done = true; break;
      case 2: LAUNCH(128, 2); done = true; break;
      case 4: LAUNCH(128, 4); done = true; break;
      case 8: LAUNCH(128, 8); done = true; break;
    }
  } else {
    switch (G) {
      case 1: LAUNCH(64, 1); done = true; break;
      case 2: LAUNCH(64, 2); done = true; break;
      case 4: LAUNCH(64, 4); done = true; break;
      case 8: LAUNCH(64, 8); done = true; break;
    }
  }
#undef LAUNCH
  TORCH_CHECK(done, "unsupported GQA group ", G);
  HIP_CHECK_KERNEL();

  const long long nhq = (long long)n * Hq;
  dim3 gridc((unsigned)((nhq + 3) / 4));
  if (D == 128)
    hipLaunchKernelGGL(ragged::ragged_combine_kernel<128>, gridc, dim3(256),
                       0, stream.stream(), pout.data_ptr<float>(),
                       pml.data_ptr<float>(),
                       reinterpret_cast<short*>(out.data_ptr()), nhq, NS);
  else
    hipLaunchKernelGGL(ragged::ragged_combine_kernel<64>, gridc, dim3(256),
                       0, stream.stream(), pout.data_ptr<float>(),
                       pml.data_ptr<float>(),
                       reinterpret_cast<short*>(out.data_ptr()), nhq, NS);
  HIP_CHECK_KERNEL();
  return out;
}
