// Flash attention forward for MI355X (gfx950) — hand-written MFMA kernel.
//
// Role parity: reference uses CUTLASS/flash-attn kernels
// (deepspeed/inference/v2/kernels, csrc/transformer/inference/csrc/softmax.cu).
// MI355X-native design (guide §B "fused attention prefill"):
//   - 4 waves/block, 64 q-rows per block (16 per wave), KV tiles of 64
//   - mfma_f32_16x16x32_bf16; online softmax in fp32 registers
//   - K tile staged in LDS with the ((row&7)<<4) XOR byte-swizzle
//     (row-major [64][128] bf16 is otherwise a 16-way bank conflict)
//   - V staged TRANSPOSED in LDS so the PV B-operand is a ds_read_b128
//   - P staged per-wave in LDS (pad 8) to convert C-layout -> A-layout
//   - causal masking; GQA via head-group indexing; saves per-row LSE for the
//     backward pass
// Correctness verified against fp32 torch reference (tests/test_ops_gpu.py,
// asymmetric random inputs per guide G9).
#include <torch/extension.h>

#include "common.h"

using bf16x8_t = __attribute__((ext_vector_type(8))) short;
using f32x4_t = __attribute__((ext_vector_type(4))) float;

#define MFMA_BF16_16x16x32 __builtin_amdgcn_mfma_f32_16x16x32_bf16

constexpr int QTILE = 128;  // q rows per block (8 waves x 16)
constexpr int KVTILE = 64;  // kv rows per tile
constexpr int DHEAD = 128;
constexpr int KPAD = 0;            // K uses swizzle, no pad
constexpr int VT_PAD = 8;          // Vt rows padded: [128][64+8]
constexpr int P_PAD = 8;           // P rows padded: [16][64+8]

// swizzle byte offset within a K row (16B granules spread across banks)
DEV_INLINE int kswz(int row, int byte_off) {
  return row * (DHEAD * 2) + (byte_off ^ ((row & 7) << 4));
}

// Vt[d][kv] byte offset: pad-8 rows spread the d-read (2-way) and the
// ((d>>5)&3)<<5 XOR spreads the 4 d-groups on the transpose write
// (was an 8-way write conflict -> ~2-way). Alignment: both terms are
// multiples of 16B for the b128 reads.
DEV_INLINE int vtswz(int d, int kv_byte) {
  return d * ((KVTILE + VT_PAD) * 2) + (kv_byte ^ (((d >> 5) & 3) << 5));
}

__launch_bounds__(512, 3)
__global__ void flash_fwd_kernel(
    const short* __restrict__ q,  // [B,S,Hq,D]
    const short* __restrict__ k,  // [B,S,Hk,D]
    const short* __restrict__ v,  // [B,S,Hk,D]
    short* __restrict__ out,      // [B,S,Hq,D]
    float* __restrict__ lse_out,  // [B,Hq,S]
    int B, int S, int Hq, int Hk, float scale, int causal) {
  __shared__ short k_lds[KVTILE * DHEAD];            // swizzled
  __shared__ short vt_lds[DHEAD * (KVTILE + VT_PAD)];  // transposed
  __shared__ short p_lds[8][16 * (KVTILE + P_PAD)];    // per-wave P

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int l15 = lane & 15;
  const int l4 = lane >> 4;  // 0..3

  const int qtile = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int h = bh % Hq;
  const int hk = h / (Hq / Hk);
  const int qbase = qtile * QTILE;
  const int wave_q = qbase + wave * 16;  // this wave's first q row

  const long long q_row_stride = (long long)Hq * DHEAD;
  const long long kv_row_stride = (long long)Hk * DHEAD;
  const short* qp = q + ((long long)b * S) * q_row_stride + h * DHEAD;
  const short* kp = k + ((long long)b * S) * kv_row_stride + hk * DHEAD;
  const short* vp = v + ((long long)b * S) * kv_row_stride + hk * DHEAD;

  // ---- load Q fragments: lane holds Q[wave_q + l15][l4*8 + j + 32*ks]
  bf16x8_t qfrag[4];
  {
    int qrow = wave_q + l15;
    const short* src = qp + (long long)qrow * q_row_stride;
    bool valid = qrow < S;
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      if (valid) {
        qfrag[ks] = *reinterpret_cast<const bf16x8_t*>(src + ks * 32 + l4 * 8);
      } else {
        qfrag[ks] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
      }
    }
  }

  // online softmax state: rows wave_q + l4*4 + r  (r = 0..3)
  float m_run[4], l_run[4];
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    m_run[r] = -INFINITY;
    l_run[r] = 0.f;
  }
  f32x4_t o_acc[8];
#pragma unroll
  for (int dt = 0; dt < 8; ++dt) o_acc[dt] = f32x4_t{0.f, 0.f, 0.f, 0.f};

  const int kv_end = causal ? min(S, qbase + QTILE) : S;
  const int n_tiles = (kv_end + KVTILE - 1) / KVTILE;

  for (int t = 0; t < n_tiles; ++t) {
    const int kvbase = t * KVTILE;
    // ---- stage K (async global_load_lds, pre-swizzled source) -----------
    // wave w, pass p covers 1 KiB of linear K LDS: lane's dest is
    // base + lane*16 (HW rule), so the SOURCE column carries the XOR
    // (guide m173: swizzled layouts via pre-swizzled global address).
    {
      for (int pass = 0; pass < 2; ++pass) {
        int linear = (pass * 8 + wave) * 1024 + lane * 16;
        int row = linear >> 8;             // /256 bytes per row
        int colbyte = linear & 255;
        int src_col = colbyte ^ ((row & 7) << 4);
        int grow = kvbase + row;
        int srow = grow < S ? grow : S - 1;  // clamped; masked via -inf
        const short* src = kp + (long long)srow * kv_row_stride +
                           (src_col >> 1);
        __builtin_amdgcn_global_load_lds(
            reinterpret_cast<const unsigned int*>(src),
            reinterpret_cast<unsigned int*>(
                reinterpret_cast<char*>(k_lds) + (pass * 8 + wave) * 1024),
            16, 0, 0);
      }
      // V transposed: coalesced row reads, XOR-spread scatter writes
      int row = threadIdx.x >> 3;        // 0..63
      int c0 = (threadIdx.x & 7) * 16;   // 0..112 step 16
      int grow = kvbase + row;
      const short* vrow = vp + (long long)grow * kv_row_stride;
#pragma unroll
      for (int cc = 0; cc < 2; ++cc) {
        int col = c0 + cc * 8;
        bf16x8_t vv8;
        if (grow < S) {
          vv8 = *reinterpret_cast<const bf16x8_t*>(vrow + col);
        } else {
          vv8 = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};
        }
        char* vbase = reinterpret_cast<char*>(vt_lds);
#pragma unroll
        for (int j = 0; j < 8; ++j)
          *reinterpret_cast<short*>(vbase + vtswz(col + j, row * 2)) =
              vv8[j];
      }
      asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
    }
    __syncthreads();

    // ---- S = Q @ K^T  (4 col-tiles of 16) -------------------------------
    f32x4_t s_acc[4];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ct = 0; ct < 4; ++ct) {
      f32x4_t acc = {0.f, 0.f, 0.f, 0.f};
      int krow = l15 + 16 * ct;
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(k_lds) + kswz(krow, (l4 * 8 + 32 * ks) * 2));
        acc = MFMA_BF16_16x16x32(qfrag[ks], bfrag, acc, 0, 0, 0);
      }
      s_acc[ct] = acc;
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- online softmax -------------------------------------------------
    // lane holds S[row = l4*4 + r][col = l15 + 16*ct] (scaled below)
    const bool btile = (causal && (kvbase + KVTILE > qbase)) ||
                       (kvbase + KVTILE > S);
    float p_new[4][4];  // [ct][r]
    float corr[4];
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      int grow = wave_q + l4 * 4 + r;
      float rowmax = -INFINITY;
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        float sv = s_acc[ct][r] * scale;
        if (btile) {
          int gcol = kvbase + l15 + 16 * ct;
          if ((causal && gcol > grow) || gcol >= S) sv = -INFINITY;
        }
        s_acc[ct][r] = sv;
        rowmax = fmaxf(rowmax, sv);
      }
      // reduce across the 16 lanes sharing these rows (DPP, VALU-only)
      rowmax = row16_reduce_max(rowmax);
      float m_new = fmaxf(m_run[r], rowmax);
      float c = (m_run[r] == -INFINITY) ? 0.f : __expf(m_run[r] - m_new);
      corr[r] = c;
      float rowsum = 0.f;
#pragma unroll
      for (int ct = 0; ct < 4; ++ct) {
        float p = (s_acc[ct][r] == -INFINITY) ? 0.f
                                              : __expf(s_acc[ct][r] - m_new);
        p_new[ct][r] = p;
        rowsum += p;
      }
      rowsum = row16_reduce_sum(rowsum);
      l_run[r] = l_run[r] * c + rowsum;
      m_run[r] = m_new;
    }
    // rescale O
#pragma unroll
    for (int dt = 0; dt < 8; ++dt)
#pragma unroll
      for (int r = 0; r < 4; ++r) o_acc[dt][r] *= corr[r];

    // ---- stage P (bf16) into per-wave LDS -------------------------------
    short* pw = p_lds[wave];
#pragma unroll
    for (int ct = 0; ct < 4; ++ct)
#pragma unroll
      for (int r = 0; r < 4; ++r)
        pw[(l4 * 4 + r) * (KVTILE + P_PAD) + l15 + 16 * ct] =
            f2bf(p_new[ct][r]);
    // wave-local P handoff: drain the ds_writes before cross-lane ds_reads
    // (no cross-wave barrier needed — each wave reads only its own region;
    // the "memory" clobber pins ordering, guide §5 rule 18)
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");

    // ---- O += P @ V  ----------------------------------------------------
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16x8_t afrag = *reinterpret_cast<const bf16x8_t*>(
          pw + l15 * (KVTILE + P_PAD) + l4 * 8 + 32 * ks);
#pragma unroll
      for (int dt = 0; dt < 8; ++dt) {
        bf16x8_t bfrag = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(vt_lds) +
            vtswz(l15 + 16 * dt, (l4 * 8 + 32 * ks) * 2));
        o_acc[dt] = MFMA_BF16_16x16x32(afrag, bfrag, o_acc[dt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();  // before next tile's staging overwrites K/V
  }

  // ---- epilogue: O /= l, write out + LSE --------------------------------
#pragma unroll
  for (int r = 0; r < 4; ++r) {
    int grow = wave_q + l4 * 4 + r;
    if (grow >= S) continue;
    float inv_l = (l_run[r] > 0.f) ? 1.f / l_run[r] : 0.f;
    short* orow = out + ((long long)b * S + grow) * q_row_stride + h * DHEAD;
#pragma unroll
    for (int dt = 0; dt < 8; ++dt)
      orow[l15 + 16 * dt] = f2bf(o_acc[dt][r] * inv_l);
    if (l15 == 0 && lse_out)
      lse_out[((long long)b * Hq + h) * S + grow] =
          m_run[r] + __logf(fmaxf(l_run[r], 1e-30f));
  }
}


// ===========================================================================
// v5: 8-wave 32x32 "swapped QK^T" forward (guide §B 8-warp ladder).
//   - each wave owns 32 q rows (block = 256 q rows), KV tiles of 64
//   - S^T = K·Q^T via mfma_f32_32x32x16_bf16: C col = q = lane&31, so the
//     softmax row (over kv) is LANE-LOCAL: in-lane reduce + one half-swap
//   - P stays in registers: cvt_pk to bf16 pairs + permlane32_swap build
//     the PV A/B fragments directly — no P LDS staging, no cross-wave sync
//   - PV computed as O^T = V^T·P^T so O's q is also lane-local and the
//     online-softmax rescale is a scalar multiply
//   - K staged via global_load_lds (pre-swizzled source), V transposed
// C/D layout (32x32x16): col=lane&31, row=(reg&3)+8*(reg>>2)+4*(lane>>5).
// A/B operand: row(col)=lane&31, k=(lane>>5)*8+j.
// ===========================================================================

using f32x16_t = __attribute__((ext_vector_type(16))) float;
#define MFMA_BF16_32x32x16 __builtin_amdgcn_mfma_f32_32x32x16_bf16

constexpr int QB5 = 32;    // q rows per wave
constexpr int QT5 = 256;   // q rows per block (8 waves)
constexpr int KT5 = 64;    // kv rows per tile

// exchange a scalar with the other half-wave (VALU permlane, no LDS pipe)
DEV_INLINE float half_swap(float x, int hi) {
  int i = __builtin_bit_cast(int, x);
  auto r = __builtin_amdgcn_permlane32_swap(i, i, false, false);
  return __builtin_bit_cast(float, hi ? r[0] : r[1]);
}

// pack two f32 into a u32 of 2 bf16 (compiler emits v_cvt_pk_bf16_f32)
DEV_INLINE unsigned int cvtpk2(float lo, float hi) {
  unsigned short a = __builtin_bit_cast(unsigned short, (__bf16)lo);
  unsigned short b = __builtin_bit_cast(unsigned short, (__bf16)hi);
  return (unsigned int)a | ((unsigned int)b << 16);
}

// swizzled row-major byte offset for a [R][DH] bf16 tile
template <int DH>
DEV_INLINE int kswz_t(int row, int byte_off) {
  return row * (DH * 2) + (byte_off ^ ((row & 7) << 4));
}

// kvmask: optional [B,S] additive float mask over kv columns (padding
// masks, ALiBi-free BERT-class workloads); nullptr = none.
template <int DH>
__launch_bounds__(512, 2)
__global__ void flash_fwd_v5_kernel(
    const short* __restrict__ q,  // [B,S,Hq,D]
    const short* __restrict__ k,  // [B,S,Hk,D]
    const short* __restrict__ v,  // [B,S,Hk,D]
    short* __restrict__ out,      // [B,S,Hq,D]
    float* __restrict__ lse_out,  // [B,Hq,S]
    const float* __restrict__ kvmask,  // [B,S] or null
    int B, int S, int Hq, int Hk, float scale, int causal) {
  // K double-buffered: tile t+1's global_load_lds issues during tile t's
  // compute (async-STAGE, guide T14); V global loads land in registers one
  // tile early and scatter to LDS after the barrier.
  __shared__ short k_lds[2][KT5 * DH];              // swizzled row-major
  __shared__ short vt_lds[DH * (KT5 + VT_PAD)];     // transposed

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int l31 = lane & 31;
  const int hi = lane >> 5;  // half-wave: 0 or 1

  constexpr int NQF = DH / 16;   // Q k-slices
  constexpr int NOT = DH / 32;   // O^T d m-tiles

  const int qtile = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int h = bh % Hq;
  const int hk = h / (Hq / Hk);
  const int qbase = qtile * QT5;
  const int wave_q = qbase + wave * QB5;
  const int my_q = wave_q + l31;  // this lane's q row (lane-local softmax)

  const long long qrs = (long long)Hq * DH;
  const long long kvrs = (long long)Hk * DH;
  const short* qp = q + ((long long)b * S) * qrs + h * DH;
  const short* kp = k + ((long long)b * S) * kvrs + hk * DH;
  const short* vp = v + ((long long)b * S) * kvrs + hk * DH;
  const float* mp = kvmask ? kvmask + (long long)b * S : nullptr;

  // ---- Q fragments: qfrag[ks] = Q[my_q][16*ks + hi*8 .. +7] -------------
  bf16x8_t qfrag[NQF];
  {
    int srow = my_q < S ? my_q : S - 1;
    const short* src = qp + (long long)srow * qrs + hi * 8;
#pragma unroll
    for (int ks = 0; ks < NQF; ++ks)
      qfrag[ks] = *reinterpret_cast<const bf16x8_t*>(src + 16 * ks);
  }

  // softmax runs in the exp2 domain (v_exp_f32 is natively 2^x): S values
  // are scaled by scale*log2(e) once, so every exp is a bare v_exp with no
  // hidden *log2e multiply; LSE converts back with ln2 at the epilogue.
  const float sc2 = scale * 1.44269504f;
  float m_run = -INFINITY, l_run = 0.f;
  f32x16_t ot[NOT];  // O^T accum: rows d = 32*mt + (reg&3)+8*(reg>>2)+4*hi
#pragma unroll
  for (int mt = 0; mt < NOT; ++mt)
#pragma unroll
    for (int r = 0; r < 16; ++r) ot[mt][r] = 0.f;

  const int kv_end = causal ? min(S, qbase + QT5) : S;
  const int n_tiles = (kv_end + KT5 - 1) / KT5;

  // async-STAGE helpers -----------------------------------------------------
  // V staging: DH/16 threads per row, 16 cols each (guard rows >= KT5 for
  // DH < 128 where 512 threads cover more than one tile)
  constexpr int VTPR = DH / 16;
  const int st_row = threadIdx.x / VTPR;
  const int st_c0 = (threadIdx.x % VTPR) * 16;
  // K tile = KT5*DH*2 bytes in 1024-byte wave segments (64 lanes x 16 B)
  constexpr int KSEGS = KT5 * DH * 2 / 1024;
  constexpr int KPASSES = (KSEGS + 7) / 8;

#define ISSUE_K(t_)                                                          \
  {                                                                          \
    int kvb_ = (t_)*KT5;                                                     \
    short* kbuf_ = k_lds[(t_)&1];                                            \
    for (int pass = 0; pass < KPASSES; ++pass) {                             \
      int seg = pass * 8 + wave;                                             \
      int linear = seg * 1024 + lane * 16;                                   \
      int row = linear / (DH * 2);                                           \
      int colbyte = linear % (DH * 2);                                       \
      int src_col = colbyte ^ ((row & 7) << 4);                              \
      int grow = kvb_ + row;                                                 \
      int srow = grow < S ? grow : S - 1;                                    \
      const short* src = kp + (long long)srow * kvrs + (src_col >> 1);       \
      __builtin_amdgcn_global_load_lds(                                      \
          reinterpret_cast<const unsigned int*>(src),                        \
          reinterpret_cast<unsigned int*>(reinterpret_cast<char*>(kbuf_) +   \
                                          seg * 1024),                       \
          16, 0, 0);                                                         \
    }                                                                        \
  }

#define LOAD_V(t_, dst_)                                                     \
  {                                                                          \
    int grow = (t_)*KT5 + st_row;                                            \
    const short* vrow = vp + (long long)grow * kvrs;                         \
    if (st_row < KT5 && grow < S) {                                          \
      dst_[0] = *reinterpret_cast<const bf16x8_t*>(vrow + st_c0);            \
      dst_[1] = *reinterpret_cast<const bf16x8_t*>(vrow + st_c0 + 8);        \
    } else {                                                                 \
      dst_[0] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};                            \
      dst_[1] = bf16x8_t{0, 0, 0, 0, 0, 0, 0, 0};                            \
    }                                                                        \
  }

  bf16x8_t vreg[2], vnext[2];
  ISSUE_K(0);
  LOAD_V(0, vreg);

  for (int t = 0; t < n_tiles; ++t) {
    const int kvbase = t * KT5;
    const bool has_next = (t + 1) < n_tiles;
    if (has_next) ISSUE_K(t + 1);  // lands in the other K buffer
    // scatter this tile's V registers into LDS (the compiler's wait for
    // vreg also retires this tile's older global_load_lds -> k_lds)
    if (st_row < KT5) {
      char* vbase = reinterpret_cast<char*>(vt_lds);
#pragma unroll
      for (int cc = 0; cc < 2; ++cc) {
        int col = st_c0 + cc * 8;
#pragma unroll
        for (int j = 0; j < 8; ++j)
          *reinterpret_cast<short*>(vbase + vtswz(col + j, st_row * 2)) =
              vreg[cc][j];
      }
    }
    if (has_next) LOAD_V(t + 1, vnext);  // in flight under this tile's MFMAs
    __syncthreads();

    // ---- S^T = K Q^T: st[mt] rows kv = kvbase+32mt+(r&3)+8*(r>>2)+4hi ---
    f32x16_t st[2];
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mt = 0; mt < 2; ++mt) {
      f32x16_t acc;
#pragma unroll
      for (int r = 0; r < 16; ++r) acc[r] = 0.f;
      int krow = 32 * mt + l31;
#pragma unroll
      for (int ks = 0; ks < NQF; ++ks) {
        bf16x8_t kf = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(k_lds[t & 1]) +
            kswz_t<DH>(krow, (16 * ks + hi * 8) * 2));
        acc = MFMA_BF16_32x32x16(kf, qfrag[ks], acc, 0, 0, 0);
      }
      st[mt] = acc;
    }
    __builtin_amdgcn_s_setprio(0);

    // ---- online softmax (lane-local q row) ------------------------------
    // btile is wave-uniform: interior tiles take the cmp-free path
    const bool btile = (causal && (kvbase + KT5 > qbase)) ||
                       (kvbase + KT5 > S) || mp != nullptr;
    float tmax = -INFINITY;
    if (btile) {
#pragma unroll
      for (int mt = 0; mt < 2; ++mt)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float sv = st[mt][r] * sc2;
          int gkv = kvbase + 32 * mt + (r & 3) + 8 * (r >> 2) + 4 * hi;
          if (mp && gkv < S) sv += mp[gkv] * 1.44269504f;
          if ((causal && gkv > my_q) || gkv >= S) sv = -INFINITY;
          st[mt][r] = sv;
          tmax = fmaxf(tmax, sv);
        }
    } else {
#pragma unroll
      for (int mt = 0; mt < 2; ++mt)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          float sv = st[mt][r] * sc2;
          st[mt][r] = sv;
          tmax = fmaxf(tmax, sv);
        }
    }
    tmax = fmaxf(tmax, half_swap(tmax, hi));
    // defer-max (guide T13): only rescale when the running max grew by more
    // than 8/ln2 (P is then bounded by e^8, which f32 accum tolerates; LSE
    // stays exact). Masked values rely on exp2(-inf)=0 — no per-value select.
    if (!__all(tmax - m_run <= 11.5416f)) {
      float m_new = fmaxf(m_run, tmax);
      float c = (m_run == -INFINITY) ? 0.f : __builtin_amdgcn_exp2f(m_run - m_new);
      l_run *= c;
#pragma unroll
      for (int mt = 0; mt < NOT; ++mt)
#pragma unroll
        for (int r = 0; r < 16; ++r) ot[mt][r] *= c;
      m_run = m_new;
    }
    float psum = 0.f;
#pragma unroll
    for (int mt = 0; mt < 2; ++mt)
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float p = __builtin_amdgcn_exp2f(st[mt][r] - m_run);
        st[mt][r] = p;  // P overwrites S in-register
        psum += p;
      }
    psum += half_swap(psum, hi);
    l_run += psum;

    // ---- build P fragments in-register ----------------------------------
    // pfrag[ks] = P[q=l31][kv = kvbase + 16ks + hi*8 + j]; own C regs hold
    // kv (r&3)+8*(r>>2)+4hi — the other half-wave holds the 4-offset rows.
    // permlane32_swap(wA, wB): x = {lo: wA_lo, hi: wB_lo},
    //                          y = {lo: wA_hi, hi: wB_hi}
    // with wA = own rows +0..3 packed, wB = own rows +4..7 packed gives
    // each half the partner words it needs (guide T12).
    bf16x8_t pfrag[4];
#pragma unroll
    for (int ks = 0; ks < 4; ++ks) {
      int mt = ks >> 1, s = ks & 1;  // kv 16s..16s+15 of m-tile mt
      // own regs 8s..8s+3 = rows 16s+4hi+0..3 ; 8s+4..8s+7 = +8
      unsigned int wA0 = cvtpk2(st[mt][8 * s + 0], st[mt][8 * s + 1]);
      unsigned int wA1 = cvtpk2(st[mt][8 * s + 2], st[mt][8 * s + 3]);
      unsigned int wB0 = cvtpk2(st[mt][8 * s + 4], st[mt][8 * s + 5]);
      unsigned int wB1 = cvtpk2(st[mt][8 * s + 6], st[mt][8 * s + 7]);
      auto r0 = __builtin_amdgcn_permlane32_swap((int)wA0, (int)wB0,
                                                 false, false);
      auto r1 = __builtin_amdgcn_permlane32_swap((int)wA1, (int)wB1,
                                                 false, false);
      unsigned int w[4] = {(unsigned int)r0[0], (unsigned int)r1[0],
                           (unsigned int)r0[1], (unsigned int)r1[1]};
      pfrag[ks] = *reinterpret_cast<bf16x8_t*>(w);
    }

    // ---- O^T += V^T P^T  (4 d m-tiles x 4 kv k-slices) ------------------
    __builtin_amdgcn_s_setprio(1);
#pragma unroll
    for (int mt = 0; mt < NOT; ++mt) {
      int drow = 32 * mt + l31;
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        bf16x8_t vf = *reinterpret_cast<const bf16x8_t*>(
            reinterpret_cast<char*>(vt_lds) +
            vtswz(drow, (16 * ks + hi * 8) * 2));
        ot[mt] = MFMA_BF16_32x32x16(vf, pfrag[ks], ot[mt], 0, 0, 0);
      }
    }
    __builtin_amdgcn_s_setprio(0);
    __syncthreads();
    vreg[0] = vnext[0];
    vreg[1] = vnext[1];
  }
#undef ISSUE_K
#undef LOAD_V

  // ---- epilogue: O^T lane holds d rows for its q col --------------------
  if (my_q < S) {
    float inv_l = (l_run > 0.f) ? 1.f / l_run : 0.f;
    short* orow = out + ((long long)b * S + my_q) * qrs + h * DH;
#pragma unroll
    for (int mt = 0; mt < NOT; ++mt)
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        int d0 = 32 * mt + 8 * g + 4 * hi;  // 4 consecutive d
        unsigned int w[2];
        w[0] = cvtpk2(ot[mt][4 * g + 0] * inv_l, ot[mt][4 * g + 1] * inv_l);
        w[1] = cvtpk2(ot[mt][4 * g + 2] * inv_l, ot[mt][4 * g + 3] * inv_l);
        *reinterpret_cast<unsigned long long*>(orow + d0) =
            *reinterpret_cast<unsigned long long*>(w);
      }
    if (hi == 0 && lse_out)  // back to base-e: lse = ln2*(m2 + log2 l)
      lse_out[((long long)b * Hq + h) * S + my_q] =
          0.693147181f * (m_run + __log2f(fmaxf(l_run, 1e-30f)));
  }
}

std::vector<at::Tensor> flash_attn_fwd(at::Tensor q, at::Tensor k,
                                       at::Tensor v, bool causal,
                                       double scale,
                                       c10::optional<at::Tensor> kvmask) {
  TORCH_CHECK(q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  int Hk = k.size(2);
  TORCH_CHECK(D == 128 || D == 64,
              "flash_attn_fwd: head_dim must be 64 or 128");
  TORCH_CHECK(Hq % Hk == 0, "GQA requires Hq % Hk == 0");
  const float* mp = nullptr;
  if (kvmask.has_value() && kvmask->defined()) {
    TORCH_CHECK(kvmask->scalar_type() == at::kFloat &&
                kvmask->is_contiguous() && kvmask->numel() == (long)B * S,
                "kvmask must be contiguous float [B,S]");
    mp = kvmask->data_ptr<float>();
  }
  auto out = at::empty_like(q);
  auto lse = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  static const bool use_v4 = []() {
    const char* e = getenv("DSAMD_ATTN_FWD_V4");
    return e && e[0] == '1';
  }();
  if (use_v4 && D == 128 && !mp) {
    dim3 grid((S + QTILE - 1) / QTILE, B * Hq);
    hipLaunchKernelGGL(flash_fwd_kernel, grid, dim3(512), 0, stream.stream(),
                       reinterpret_cast<const short*>(q.data_ptr()),
                       reinterpret_cast<const short*>(k.data_ptr()),
                       reinterpret_cast<const short*>(v.data_ptr()),
                       reinterpret_cast<short*>(out.data_ptr()),
                       lse.data_ptr<float>(), B, S, Hq, Hk, (float)scale,
                       causal ? 1 : 0);
  } else {
    dim3 grid((S + QT5 - 1) / QT5, B * Hq);
    if (D == 128) {
      hipLaunchKernelGGL(flash_fwd_v5_kernel<128>, grid, dim3(512), 0,
                         stream.stream(),
                         reinterpret_cast<const short*>(q.data_ptr()),
                         reinterpret_cast<const short*>(k.data_ptr()),
                         reinterpret_cast<const short*>(v.data_ptr()),
                         reinterpret_cast<short*>(out.data_ptr()),
                         lse.data_ptr<float>(), mp, B, S, Hq, Hk,
                         (float)scale, causal ? 1 : 0);
    } else {
      hipLaunchKernelGGL(flash_fwd_v5_kernel<64>, grid, dim3(512), 0,
                         stream.stream(),
                         reinterpret_cast<const short*>(q.data_ptr()),
                         reinterpret_cast<const short*>(k.data_ptr()),
                         reinterpret_cast<const short*>(v.data_ptr()),
                         reinterpret_cast<short*>(out.data_ptr()),
                         lse.data_ptr<float>(), mp, B, S, Hq, Hk,
                         (float)scale, causal ? 1 : 0);
    }
  }
  HIP_CHECK_KERNEL();
  return {out, lse};
}
