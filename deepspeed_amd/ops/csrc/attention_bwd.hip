// Flash attention BACKWARD for MI355X (gfx950) — hand-written MFMA kernels.
//
// Exact flash backward (recompute P from saved LSE), two passes:
//   pass 1 (kv-outer): dV = P^T dO ; dK = dS^T Q    — no atomics
//   pass 2 (q-outer):  dQ = dS K                     — no atomics
// with dS = P o (dP - Drow) * scale, dP = dO V^T, P = exp(S*scale - lse).
// Drow = rowsum(dO o O) is precomputed by the wrapper.
//
// Layout conventions follow the forward kernel (attention.hip): C/D layout
// row=(lane>>4)*4+reg, col=lane&15; A-operand lane holds A[l15][l4*8+j];
// row-major LDS tiles use the ((row&7)<<4) XOR swizzle with pre-swizzled
// staging; transposed tiles use pad-8 rows + a 16B-granule XOR.
// GQA: pass 1 accumulates over the q-head group in-kernel (K/V fragments
// stay in registers across the group), writing dK/dV at [B,S,Hk,D]
// directly — no per-q-head buffers, no wrapper reduction.
#include <torch/extension.h>

#include "common.h"

using bf16x8_t = __attribute__((ext_vector_type(8))) short;
using f32x4_t = __attribute__((ext_vector_type(4))) float;

#define MFMA_BF16_16x16x32 __builtin_amdgcn_mfma_f32_16x16x32_bf16

namespace bwd {

// drow[b,h,s] = sum_d dout[b,s,h,d] * out[b,s,h,d]  (one wave per row)
template <int DH>
__global__ void drow_kernel(const short* __restrict__ dout,
                            const short* __restrict__ out,
                            float* __restrict__ drow, int S, int Hq,
                            long long n_rows) {
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  long long row = (long long)blockIdx.x * 4 + wave;  // over B*S*Hq
  if (row >= n_rows) return;
  const long long nbsh = (long long)S * Hq;
  const short* a = dout + row * DH;
  const short* b = out + row * DH;
  // DH/2 u32 words over 64 lanes (1 word/lane at DH=128)
  float acc = 0.f;
  for (int w = lane; w < DH / 2; w += 64) {
    unsigned int au = reinterpret_cast<const unsigned int*>(a)[w];
    unsigned int bu = reinterpret_cast<const unsigned int*>(b)[w];
    acc += bf2f((short)(au & 0xffff)) * bf2f((short)(bu & 0xffff)) +
           bf2f((short)(au >> 16)) * bf2f((short)(bu >> 16));
  }
  acc = wave_reduce_sum(acc);
  if (lane == 0) {
    // row index [b, s, h] -> drow[b, h, s]
    long long bb = row / nbsh;
    long long rem = row % nbsh;
    long long s = rem / Hq;
    long long h = rem % Hq;
    drow[(bb * Hq + h) * S + s] = acc;
  }
}

// row-major [R][DH] bf16 with ((row&7)<<4) XOR
template <int DH>
DEV_INLINE int rmswz(int row, int byte_off) {
  return row * (DH * 2) + (byte_off ^ ((row & 7) << 4));
}

// transposed [128][W + 8] bf16 with 16B-granule XOR on the inner index
template <int W>
DEV_INLINE int trswz(int d, int inner_byte) {
  return d * ((W + 8) * 2) + (inner_byte ^ (((d >> 5) & 3) << 4));
}

DEV_INLINE short f2bf_(float f) { return f2bf(f); }

// ---------------------------------------------------------------------------
// PASS 1: kv-outer. Block: 4 waves x 16 kv rows = 64 kv rows; iterates q
// tiles of 32. Computes per-(q-head) dK, dV.
// ---------------------------------------------------------------------------
template <int DH>
__launch_bounds__(256, 2)
__global__ void flash_bwd_dkv_kernel(
    const short* __restrict__ q,     // [B,S,Hq,D]
    const short* __restrict__ k,     // [B,S,Hk,D]
    const short* __restrict__ v,     // [B,S,Hk,D]
    const short* __restrict__ dout,  // [B,S,Hq,D]
    const float* __restrict__ lse,   // [B,Hq,S]
    const float* __restrict__ drow,  // [B,Hq,S]
    const float* __restrict__ kvmask,  // [B,S] or null
    short* __restrict__ dk,          // [B,S,Hk,D]
    short* __restrict__ dv,          // [B,S,Hk,D]
    int B, int S, int Hq, int Hk, float scale, int causal) {
  constexpr int QIT = 32;
  constexpr int NKS = DH / 32;  // A-operand k-slices
  constexpr int NDT = DH / 16;  // d-tiles
  __shared__ short q_lds[QIT * DH];               // row-major swz
  __shared__ short do_lds[QIT * DH];              // row-major swz
  __shared__ short qt_lds[DH * (QIT + 8)];        // transposed
  __shared__ short dot_lds[DH * (QIT + 8)];       // transposed
  __shared__ short pw_lds[4][16 * (QIT + 8)];        // per-wave P^T stage
  __shared__ short dsw_lds[4][16 * (QIT + 8)];       // per-wave dS^T stage

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const int kv_block = blockIdx.x;      // 64 kv rows per block
  const int bh = blockIdx.y;            // over B * Hk (kv heads)
  const int b = bh / Hk;
  const int hk = bh % Hk;
  const int G = Hq / Hk;
  const int kv_base = kv_block * 64;
  const int wave_kv = kv_base + wave * 16;

  const long long qrs = (long long)Hq * DH;
  const long long kvrs = (long long)Hk * DH;
  const short* kp = k + ((long long)b * S) * kvrs + hk * DH;
  const short* vp = v + ((long long)b * S) * kvrs + hk * DH;
  const float* mp = kvmask ? kvmask + (long long)b * S : nullptr;

  // K/V fragments in registers (A-operand layout, DH/32 d-slices)
  bf16x8_t kfrag[NKS], vfrag[NKS];
  {
    int krow = wave_kv + l15;
    int srow = krow < S ? krow : S - 1;
    const short* ks = kp + (long long)srow * kvrs;
    const short* vs = vp + (long long)srow * kvrs;
#pragma unroll
    for (int ks_i = 0; ks_i < NKS; ++ks_i) {
      kfrag[ks_i] = *reinterpret_cast<const bf16x8_t*>(ks + ks_i * 32 + l4 * 8);
      vfrag[ks_i] = *reinterpret_cast<const bf16x8_t*>(vs + ks_i * 32 + l4 * 8);
    }
  }

  f32x4_t dk_acc[NDT], dv_acc[NDT];
#pragma unroll
  for (int dt = 0; dt < NDT; ++dt) {
    dk_acc[dt] = f32x4_t{0.f, 0.f, 0.f, 0.f};
    dv_acc[dt] = f32x4_t{0.f, 0.f, 0.f, 0.f};
  }

  const int q_start = causal ? (kv_base / QIT) * QIT : 0;
  // accumulate over the GQA group: all q-heads sharing this kv head
  for (int g = 0; g < G; ++g) {
  const int h = hk * G + g;
  const short* qp = q + ((long long)b * S) * qrs + h * DH;
  const short* dop = dout + ((long long)b * S) * qrs + h * DH;
  const float* lsep = lse + ((long long)b * Hq + h) * S;
  const float* drp = drow + ((long long)b * Hq + h) * S;
  for (int qb = q_start; qb < S; qb += QIT) {
    // ---- stage Q/dO (row-major swizzled + transposed) ------------------
    {
      // DH/16 threads per row, 16 cols each (2 x bf16x8)
      constexpr int TPR = DH / 16;
      int row = threadIdx.x / TPR;
      int c0 = (threadIdx.x % TPR) * 16;
      if (row < QIT) {
        int grow = qb + row;
        int srow = grow < S ? grow : S - 1;
        const short* qs = qp + (long long)srow * qrs;
        const short* ds = dop + (long long)srow * qrs;
        char* qb_ = reinterpret_cast<char*>(q_lds);
        char* db_ = reinterpret_cast<char*>(do_lds);
        char* qtb = reinterpret_cast<char*>(qt_lds);
        char* dtb = reinterpret_cast<char*>(dot_lds);
#pragma unroll
        for (int cc = 0; cc < 2; ++cc) {
          int col = c0 + cc * 8;
          bf16x8_t qv = *reinterpret_cast<const bf16x8_t*>(qs + col);
          bf16x8_t dv = *reinterpret_cast<const bf16x8_t*>(ds + col);
          *reinterpret_cast<bf16x8_t*>(qb_ + rmswz<DH>(row, col * 2)) = qv;
          *reinterpret_cast<bf16x8_t*>(db_ + rmswz<DH>(row, col * 2)) = dv;
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            *reinterpret_cast<short*>(qtb + trswz<QIT>(col + j, row * 2)) =
                qv[j];
            *reinterpret_cast<short*>(dtb + trswz<QIT>(col + j, row * 2)) =
                dv[j];
          }
        }
      }
    }
    __syncthreads();

    // ---- S^T = K Q^T  (strip: 16 kv x 32 q; 2 q-coltiles) --------------
    f32x4_t st[2], dpt[2];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ct = 0; ct < 2; ++ct) {
      f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
      f32x4_t acc2 = {0.f, 0.f, 0.f, 0.f};
      int qrow = l15 + 16 * ct;
#pragma unroll
      for (int ks_i = 0; ks_i < NKS; ++ks_i) {
        bf16x8_t bq = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(q_lds) +
            rmswz<DH>(qrow, (l4 * 8 + 32 * ks_i) * 2));
        bf16x8_t bd = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(do_lds) +
            rmswz<DH>(qrow, (l4 * 8 + 32 * ks_i) * 2));
        acc = MFMA_BF16_16x16x32(kfrag[ks_i], bq, acc, 0, 0, 0);
        acc2 = MFMA_BF16_16x16x32(vfrag[ks_i], bd, acc2, 0, 0, 0);
      }
      st[ct] = acc;    // S^T[kv][q]
      dpt[ct] = acc2;  // dP^T[kv][q] = (dO V^T)^T = V dO^T
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- P^T, dS^T ------------------------------------------------------
    short* pw = pw_lds[wave];
    short* dw = dsw_lds[wave];
#pragma unroll
    for (int ct = 0; ct < 2; ++ct) {
      int gq = qb + l15 + 16 * ct;
      float l = (gq < S) ? lsep[gq] : 0.f;
      float dr = (gq < S) ? drp[gq] : 0.f;
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int gkv = wave_kv + l4 * 4 + r;
        bool valid = gq < S && gkv < S && (!causal || gq >= gkv);
        float sval = st[ct][r] * scale;
        if (mp && gkv < S) sval += mp[gkv];
        float p = valid ? __expf(sval - l) : 0.f;
        float dsv = p * (dpt[ct][r] - dr) * scale;
        pw[(l4 * 4 + r) * (QIT + 8) + l15 + 16 * ct] = f2bf_(p);
        dw[(l4 * 4 + r) * (QIT + 8) + l15 + 16 * ct] = f2bf_(dsv);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- dV += P^T dO ; dK += dS^T Q  (8 d-tiles each) ------------------
    bf16x8_t pa = *reinterpret_cast<const bf16x8_t*>(
        pw + l15 * (QIT + 8) + l4 * 8);
    bf16x8_t da = *reinterpret_cast<const bf16x8_t*>(
        dw + l15 * (QIT + 8) + l4 * 8);
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int dt = 0; dt < NDT; ++dt) {
      bf16x8_t bdo = *reinterpret_cast<const bf16x8_t*>(
          reinterpret_cast<char*>(dot_lds) +
          trswz<QIT>(l15 + 16 * dt, (l4 * 8) * 2));
      bf16x8_t bq = *reinterpret_cast<const bf16x8_t*>(
          reinterpret_cast<char*>(qt_lds) +
          trswz<QIT>(l15 + 16 * dt, (l4 * 8) * 2));
      dv_acc[dt] = MFMA_BF16_16x16x32(pa, bdo, dv_acc[dt], 0, 0, 0);
      dk_acc[dt] = MFMA_BF16_16x16x32(da, bq, dk_acc[dt], 0, 0, 0);
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
  }
  }  // group loop

  // ---- epilogue: write dK/dV [B,S,Hk,D] ---------------------------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int gkv = wave_kv + l4 * 4 + r;
    if (gkv >= S) continue;
    short* dkr = dk + ((long long)b * S + gkv) * kvrs + hk * DH;
    short* dvr = dv + ((long long)b * S + gkv) * kvrs + hk * DH;
#pragma unroll
    for (int dt = 0; dt < NDT; ++dt) {
      dkr[l15 + 16 * dt] = f2bf_(dk_acc[dt][r]);
      dvr[l15 + 16 * dt] = f2bf_(dv_acc[dt][r]);
    }
  }
}

// ---------------------------------------------------------------------------
// PASS 2: q-outer. Block: 4 waves x 16 q rows = 64; iterates kv tiles of 64.
// ---------------------------------------------------------------------------
template <int DH>
__launch_bounds__(256, 2)
__global__ void flash_bwd_dq_kernel(
    const short* __restrict__ q, const short* __restrict__ k,
    const short* __restrict__ v, const short* __restrict__ dout,
    const float* __restrict__ lse, const float* __restrict__ drow,
    const float* __restrict__ kvmask,  // [B,S] or null
    short* __restrict__ dq,  // [B,S,Hq,D]
    int B, int S, int Hq, int Hk, float scale, int causal) {
  constexpr int KVT = 64;
  constexpr int NKS = DH / 32;
  constexpr int NDT = DH / 16;
  __shared__ short k_lds[KVT * DH];          // row-major swz
  __shared__ short v_lds[KVT * DH];          // row-major swz
  __shared__ short kt_lds[DH * (KVT + 8)];   // transposed
  __shared__ short dsw_lds[4][16 * (KVT + 8)];  // per-wave dS stage

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;

  const int qtile = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int h = bh % Hq;
  const int hk = h / (Hq / Hk);
  const int qbase = qtile * 64;
  const int wave_q = qbase + wave * 16;

  const long long qrs = (long long)Hq * DH;
  const long long kvrs = (long long)Hk * DH;
  const short* qp = q + ((long long)b * S) * qrs + h * DH;
  const short* dop = dout + ((long long)b * S) * qrs + h * DH;
  const short* kp = k + ((long long)b * S) * kvrs + hk * DH;
  const short* vp = v + ((long long)b * S) * kvrs + hk * DH;
  const float* lsep = lse + ((long long)b * Hq + h) * S;
  const float* drp = drow + ((long long)b * Hq + h) * S;
  const float* mp = kvmask ? kvmask + (long long)b * S : nullptr;

  // Q and dO fragments in registers
  bf16x8_t qfrag[NKS], dofrag[NKS];
  {
    int qrow = wave_q + l15;
    int srow = qrow < S ? qrow : S - 1;
    const short* qs = qp + (long long)srow * qrs;
    const short* ds = dop + (long long)srow * qrs;
#pragma unroll
    for (int ks_i = 0; ks_i < NKS; ++ks_i) {
      qfrag[ks_i] = *reinterpret_cast<const bf16x8_t*>(qs + ks_i * 32 + l4 * 8);
      dofrag[ks_i] =
          *reinterpret_cast<const bf16x8_t*>(ds + ks_i * 32 + l4 * 8);
    }
  }
  float lse_r[4], dr_r[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int grow = wave_q + l4 * 4 + r;
    lse_r[r] = grow < S ? lsep[grow] : 0.f;
    dr_r[r] = grow < S ? drp[grow] : 0.f;
  }

  f32x4_t dq_acc[NDT];
#pragma unroll
  for (int dt = 0; dt < NDT; ++dt) dq_acc[dt] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? min(S, qbase + 64) : S;
  const int n_tiles = (kv_end + KVT - 1) / KVT;
  for (int t = 0; t < n_tiles; ++t) {
    const int kvb = t * KVT;
    // ---- stage K/V row-major + Kt transposed ----------------------------
    {
      // DH/32 threads per row, 32 cols each (4 x bf16x8)
      constexpr int TPR = DH / 32;
      int row = threadIdx.x / TPR;
      int c0 = (threadIdx.x % TPR) * 32;
      if (row < KVT) {
        int grow = kvb + row;
        int srow = grow < S ? grow : S - 1;
        const short* ks = kp + (long long)srow * kvrs;
        const short* vs = vp + (long long)srow * kvrs;
        char* kb_ = reinterpret_cast<char*>(k_lds);
        char* vb_ = reinterpret_cast<char*>(v_lds);
        char* ktb = reinterpret_cast<char*>(kt_lds);
#pragma unroll
        for (int cc = 0; cc < 4; ++cc) {
          int col = c0 + cc * 8;
          bf16x8_t kv8 = *reinterpret_cast<const bf16x8_t*>(ks + col);
          bf16x8_t vv8 = *reinterpret_cast<const bf16x8_t*>(vs + col);
          *reinterpret_cast<bf16x8_t*>(kb_ + rmswz<DH>(row, col * 2)) = kv8;
          *reinterpret_cast<bf16x8_t*>(vb_ + rmswz<DH>(row, col * 2)) = vv8;
#pragma unroll
          for (int j = 0; j < 8; ++j)
            *reinterpret_cast<short*>(ktb + trswz<KVT>(col + j, row * 2)) =
                kv8[j];
        }
      }
    }
    __syncthreads();

    // ---- S = Q K^T ; dP = dO V^T  (4 kv-coltiles) -----------------------
    f32x4_t s_acc[4], dp_acc[4];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
      f32x4_t a1 = {0.f, 0.f, 0.f, 0.f};
      f32x4_t a2 = {0.f, 0.f, 0.f, 0.f};
      int krow = l15 + 16 * ct;
#pragma unroll
      for (int ks_i = 0; ks_i < NKS; ++ks_i) {
        bf16x8_t bk = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(k_lds) +
            rmswz<DH>(krow, (l4 * 8 + 32 * ks_i) * 2));
        bf16x8_t bv = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(v_lds) +
            rmswz<DH>(krow, (l4 * 8 + 32 * ks_i) * 2));
        a1 = MFMA_BF16_16x16x32(qfrag[ks_i], bk, a1, 0, 0, 0);
        a2 = MFMA_BF16_16x16x32(dofrag[ks_i], bv, a2, 0, 0, 0);
      }
      s_acc[ct] = a1;
      dp_acc[ct] = a2;
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- dS = P o (dP - Drow) * scale -----------------------------------
    short* dw = dsw_lds[wave];
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        int grow = wave_q + l4 * 4 + r;
        int gcol = kvb + l15 + 16 * ct;
        bool valid = grow < S && gcol < S && (!causal || gcol <= grow);
        float sval = s_acc[ct][r] * scale;
        if (mp && gcol < S) sval += mp[gcol];
        float p = valid ? __expf(sval - lse_r[r]) : 0.f;
        float dsv = p * (dp_acc[ct][r] - dr_r[r]) * scale;
        dw[(l4 * 4 + r) * (KVT + 8) + l15 + 16 * ct] = f2bf_(dsv);
      }
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- dQ += dS K  (2 kv-slices x 8 d-tiles) --------------------------
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks_i = 0; ks_i < 2; ++ks_i) {
      bf16x8_t a = *reinterpret_cast<const bf16x8_t*>(
          dw + l15 * (KVT + 8) + l4 * 8 + 32 * ks_i);
#pragma unroll
      for (int dt = 0; dt < NDT; ++dt) {
        bf16x8_t bkt = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(kt_lds) +
            trswz<KVT>(l15 + 16 * dt, (l4 * 8 + 32 * ks_i) * 2));
        dq_acc[dt] = MFMA_BF16_16x16x32(a, bkt, dq_acc[dt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
  }

  // ---- epilogue ---------------------------------------------------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int grow = wave_q + l4 * 4 + r;
    if (grow >= S) continue;
    short* dqr = dq + ((long long)b * S + grow) * qrs + h * DH;
#pragma unroll
    for (int dt = 0; dt < NDT; ++dt)
      dqr[l15 + 16 * dt] = f2bf_(dq_acc[dt][r]);
  }
}

}  // namespace bwd

template <int DH>
static void launch_bwd(at::Tensor& q, at::Tensor& k, at::Tensor& v,
                       at::Tensor& dout, at::Tensor& out, at::Tensor& lse,
                       at::Tensor& dq, at::Tensor& dk, at::Tensor& dv,
                       const float* mp, int B, int S, int Hq, int Hk,
                       float scale, int causal) {
  auto stream = c10::hip::getCurrentHIPStream();
  auto drow = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  long long rows = (long long)B * S * Hq;
  dim3 gridd((rows + 3) / 4);
  hipLaunchKernelGGL(bwd::drow_kernel<DH>, gridd, dim3(256), 0,
                     stream.stream(),
                     reinterpret_cast<const short*>(dout.data_ptr()),
                     reinterpret_cast<const short*>(out.data_ptr()),
                     drow.data_ptr<float>(), S, Hq, rows);
  HIP_CHECK_KERNEL();
  dim3 grid1((S + 63) / 64, B * Hk);
  hipLaunchKernelGGL(bwd::flash_bwd_dkv_kernel<DH>, grid1, dim3(256), 0,
                     stream.stream(),
                     reinterpret_cast<const short*>(q.data_ptr()),
                     reinterpret_cast<const short*>(k.data_ptr()),
                     reinterpret_cast<const short*>(v.data_ptr()),
                     reinterpret_cast<const short*>(dout.data_ptr()),
                     lse.data_ptr<float>(), drow.data_ptr<float>(), mp,
                     reinterpret_cast<short*>(dk.data_ptr()),
                     reinterpret_cast<short*>(dv.data_ptr()), B, S, Hq,
                     Hk, scale, causal);
  HIP_CHECK_KERNEL();
  dim3 grid2((S + 63) / 64, B * Hq);
  hipLaunchKernelGGL(bwd::flash_bwd_dq_kernel<DH>, grid2, dim3(256), 0,
                     stream.stream(),
                     reinterpret_cast<const short*>(q.data_ptr()),
                     reinterpret_cast<const short*>(k.data_ptr()),
                     reinterpret_cast<const short*>(v.data_ptr()),
                     reinterpret_cast<const short*>(dout.data_ptr()),
                     lse.data_ptr<float>(), drow.data_ptr<float>(), mp,
                     reinterpret_cast<short*>(dq.data_ptr()), B, S, Hq, Hk,
                     scale, causal);
  HIP_CHECK_KERNEL();
}

std::vector<at::Tensor> flash_attn_bwd(at::Tensor q, at::Tensor k,
                                       at::Tensor v, at::Tensor dout,
                                       at::Tensor out, at::Tensor lse,
                                       bool causal, double scale,
                                       c10::optional<at::Tensor> kvmask) {
  TORCH_CHECK(q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous() &&
              dout.is_contiguous() && out.is_contiguous());
  int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  int Hk = k.size(2);
  TORCH_CHECK(D == 128 || D == 64,
              "flash_attn_bwd: head_dim must be 64 or 128");
  const float* mp = nullptr;
  if (kvmask.has_value() && kvmask->defined()) {
    TORCH_CHECK(kvmask->scalar_type() == at::kFloat &&
                kvmask->is_contiguous() && kvmask->numel() == (long)B * S,
                "kvmask must be contiguous float [B,S]");
    mp = kvmask->data_ptr<float>();
  }
  auto dq = at::empty_like(q);
  auto dk = at::empty_like(k);
  auto dv = at::empty_like(v);
  if (D == 128)
    launch_bwd<128>(q, k, v, dout, out, lse, dq, dk, dv, mp, B, S, Hq, Hk,
                    (float)scale, causal ? 1 : 0);
  else
    launch_bwd<64>(q, k, v, dout, out, lse, dq, dk, dv, mp, B, S, Hq, Hk,
                   (float)scale, causal ? 1 : 0);
  return {dq, dk, dv};
}
