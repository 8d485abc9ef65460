// Standalone probe: validate the 8x[32][16]-subtiled LDS layout that lets
// ds_read_b64_tr_b16 deliver V^T fragments for the PV MFMA directly
// (guide T10; HK/AITER attn use 1 tr_read per MFMA). Compile:
//   hipcc --offload-arch=gfx950 tools/trread_probe.hip -o /tmp/trprobe
// Run prints PASS/FAIL per (mt,ks) fragment mapping.
//
// Layout under test (derived for MFMA_32x32x16 A-operand, DH=128,KT=64):
//   element V[kv][d] stored at
//     base(mt,ks,half) + d15 + 16*j + 64*b + 128*hi
//   with ks=kv>>4, hi=(kv>>3)&1, half=(kv>>2)&1, j=kv&3,
//        mt=d>>5,  b=(d>>4)&1,  d15=d&15,
//        base = ((mt*4+ks)*2+half)*256 elems.
// tr_read at uniform base delivers to lane l, elem j:
//   lds[base + (l&15) + j*16 + ((l>>4)&1)*64 + (l>>5)*128]
// which must equal V^T[d=32mt+(l&31)][kv=16ks+8*(l>>5)+4*half+j].
#include <hip/hip_runtime.h>

#include <cstdio>
#include <vector>

using u32x2 = __attribute__((ext_vector_type(2))) unsigned int;

constexpr int KT = 64, DH = 128;

__global__ void probe(const short* __restrict__ V, short* __restrict__ out) {
  __shared__ short lds[KT * DH];
  for (int idx = threadIdx.x; idx < KT * DH; idx += blockDim.x) {
    int kv = idx / DH, d = idx % DH;
    int ks = kv >> 4, hi = (kv >> 3) & 1, half = (kv >> 2) & 1, j = kv & 3;
    int mt = d >> 5, b = (d >> 4) & 1, d15 = d & 15;
    int addr = ((mt * 4 + ks) * 2 + half) * 256 +
               d15 + 16 * j + 64 * b + 128 * hi;
    lds[addr] = V[idx];
  }
  __syncthreads();
  int lane = threadIdx.x & 63;
  if (threadIdx.x >= 64) return;
  for (int mt = 0; mt < 4; ++mt)
    for (int ks = 0; ks < 4; ++ks)
      for (int half = 0; half < 2; ++half) {
        unsigned lds_off =
            (unsigned)(((mt * 4 + ks) * 2 + half) * 256 * 2);  // bytes
        u32x2 r;
        asm volatile(
            "ds_read_b64_tr_b16 %0, %1\n\t"
            "s_waitcnt lgkmcnt(0)"
            : "=v"(r)
            : "v"(lds_off));
        // 4 bf16 elems j=0..3 for this lane
        long long o = ((((long long)mt * 4 + ks) * 64 + lane) * 8) + half * 4;
        short* dst = out + o;
        dst[0] = (short)(r[0] & 0xffff);
        dst[1] = (short)(r[0] >> 16);
        dst[2] = (short)(r[1] & 0xffff);
        dst[3] = (short)(r[1] >> 16);
      }
}

int main() {
  std::vector<short> hV(KT * DH);
  for (int i = 0; i < KT * DH; ++i) hV[i] = (short)(i * 2654435761u >> 9);
  short *dV, *dOut;
  hipMalloc(&dV, hV.size() * 2);
  hipMalloc(&dOut, 4 * 4 * 64 * 8 * 2);
  hipMemcpy(dV, hV.data(), hV.size() * 2, hipMemcpyHostToDevice);
  hipLaunchKernelGGL(probe, dim3(1), dim3(256), 0, 0, dV, dOut);
  hipError_t e = hipDeviceSynchronize();
  if (e != hipSuccess) {
    printf("KERNEL ERROR: %s\n", hipGetErrorString(e));
    return 2;
  }
  std::vector<short> hO(4 * 4 * 64 * 8);
  hipMemcpy(hO.data(), dOut, hO.size() * 2, hipMemcpyDeviceToHost);
  int bad = 0, checked = 0;
  for (int mt = 0; mt < 4; ++mt)
    for (int ks = 0; ks < 4; ++ks)
      for (int lane = 0; lane < 64; ++lane)
        for (int jj = 0; jj < 8; ++jj) {
          int hi = lane >> 5;
          int half = jj >> 2, j = jj & 3;
          int kv = 16 * ks + 8 * hi + 4 * half + j;
          int d = 32 * mt + (lane & 31);
          short want = hV[kv * DH + d];
          short got = hO[(((mt * 4 + ks) * 64 + lane) * 8) + jj];
          checked++;
          if (want != got && bad < 10) {
            printf("MISMATCH mt%d ks%d lane%d jj%d: kv%d d%d want %d got "
                   "%d\n", mt, ks, lane, jj, kv, d, want, got);
            bad++;
          } else if (want != got) {
            bad++;
          }
        }
  printf(bad ? "FAIL: %d/%d mismatches\n" : "PASS: %d fragments verified\n",
         bad ? bad : checked, checked);
  return bad ? 1 : 0;
}
