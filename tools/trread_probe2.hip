// tr_read mapping discovery: fill LDS with identity shorts (value = elem
// index), issue ds_read_b64_tr_b16 under several per-lane addressing
// schemes, dump what every lane receives. The dump REVEALS the true
// (lane, elem) -> lds-offset mapping so the attention V layout can be
// derived from measurement instead of guesswork.
//   hipcc --offload-arch=gfx950 tools/trread_probe2.hip -o /tmp/trprobe2
#include <hip/hip_runtime.h>

#include <cstdio>
#include <vector>

using u32x2 = __attribute__((ext_vector_type(2))) unsigned int;

constexpr int N = 4096;

__global__ void probe(short* __restrict__ out) {
  __shared__ short lds[N];
  for (int i = threadIdx.x; i < N; i += blockDim.x) lds[i] = (short)i;
  __syncthreads();
  if (threadIdx.x >= 64) return;
  int lane = threadIdx.x;
  // scheme 0: per-lane addr = lane * 8 bytes (consecutive 64-bit)
  // scheme 1: uniform 0
  // scheme 2: per-lane addr = lane * 4 bytes
  // scheme 3: per-lane addr = (lane & 15) * 8 + (lane >> 4) * 512 bytes
  unsigned addrs[4] = {(unsigned)(lane * 8), 0u, (unsigned)(lane * 4),
                       (unsigned)((lane & 15) * 8 + (lane >> 4) * 512)};
  for (int s = 0; s < 4; ++s) {
    u32x2 r;
    asm volatile(
        "ds_read_b64_tr_b16 %0, %1\n\t"
        "s_waitcnt lgkmcnt(0)"
        : "=v"(r)
        : "v"(addrs[s]));
    short* dst = out + (s * 64 + lane) * 4;
    dst[0] = (short)(r[0] & 0xffff);
    dst[1] = (short)(r[0] >> 16);
    dst[2] = (short)(r[1] & 0xffff);
    dst[3] = (short)(r[1] >> 16);
  }
}

int main() {
  short* dOut;
  (void)hipMalloc(&dOut, 4 * 64 * 4 * 2);
  hipLaunchKernelGGL(probe, dim3(1), dim3(256), 0, 0, dOut);
  hipError_t e = hipDeviceSynchronize();
  if (e != hipSuccess) {
    printf("KERNEL ERROR: %s\n", hipGetErrorString(e));
    return 2;
  }
  std::vector<short> h(4 * 64 * 4);
  (void)hipMemcpy(h.data(), dOut, h.size() * 2, hipMemcpyDeviceToHost);
  const char* names[4] = {"lane*8B", "uniform0", "lane*4B",
                          "(l&15)*8B+(l>>4)*512B"};
  for (int s = 0; s < 4; ++s) {
    printf("== scheme %d (%s): lane: elems j0 j1 j2 j3 (lds indices)\n", s,
           names[s]);
    for (int l = 0; l < 64; ++l) {
      printf("l%02d: %4d %4d %4d %4d%s", l, h[(s * 64 + l) * 4],
             h[(s * 64 + l) * 4 + 1], h[(s * 64 + l) * 4 + 2],
             h[(s * 64 + l) * 4 + 3], (l % 2) ? "\n" : "   ");
    }
  }
  return 0;
}
