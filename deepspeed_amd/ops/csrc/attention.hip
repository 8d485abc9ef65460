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
      // reduce across the 16 lanes sharing these rows
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        rowmax = fmaxf(rowmax, __shfl_xor(rowmax, off, 64));
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
#pragma unroll
      for (int off = 8; off > 0; off >>= 1)
        rowsum += __shfl_xor(rowsum, off, 64);
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

std::vector<at::Tensor> flash_attn_fwd(at::Tensor q, at::Tensor k,
                                       at::Tensor v, bool causal,
                                       double scale) {
  TORCH_CHECK(q.scalar_type() == at::kBFloat16);
  TORCH_CHECK(q.is_contiguous() && k.is_contiguous() && v.is_contiguous());
  int B = q.size(0), S = q.size(1), Hq = q.size(2), D = q.size(3);
  int Hk = k.size(2);
  TORCH_CHECK(D == 128, "flash_attn_fwd: head_dim must be 128");
  TORCH_CHECK(Hq % Hk == 0, "GQA requires Hq % Hk == 0");
  auto out = at::empty_like(q);
  auto lse = at::empty({B, Hq, S}, q.options().dtype(at::kFloat));
  auto stream = c10::hip::getCurrentHIPStream();
  dim3 grid((S + QTILE - 1) / QTILE, B * Hq);
  hipLaunchKernelGGL(flash_fwd_kernel, grid, dim3(512), 0, stream.stream(),
                     reinterpret_cast<const short*>(q.data_ptr()),
                     reinterpret_cast<const short*>(k.data_ptr()),
                     reinterpret_cast<const short*>(v.data_ptr()),
                     reinterpret_cast<short*>(out.data_ptr()),
                     lse.data_ptr<float>(), B, S, Hq, Hk, (float)scale,
                     causal ? 1 : 0);
  HIP_CHECK_KERNEL();
  return {out, lse};
}
