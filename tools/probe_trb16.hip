// Empirical semantics probe for gfx950 ds_read_b64_tr_b16.
//   hipcc --offload-arch=gfx950 -O3 tools/probe_trb16.hip -o /tmp/ptr
// Fills LDS with u16 pattern lds[i] = i, has every lane of one wave
// issue the transpose read at a configurable per-lane address, dumps
// the 4 returned u16 per lane.  The printed (lane -> source offsets)
// table defines the layout contract for MFMA fragment feeding.
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void probe(unsigned short* out, int addr_mode) {
  __shared__ unsigned short lds[4096];
  const int tid = threadIdx.x;
  for (int i = tid; i < 4096; i += blockDim.x) lds[i] = (unsigned short)i;
  __syncthreads();
  if (tid >= 64) return;
  // per-lane byte address into LDS (8-B aligned)
  unsigned base = (unsigned)(unsigned long long)&lds[0];
  unsigned off;
  const int gl = tid & 15, a4 = gl >> 2, c4 = gl & 3;
  switch (addr_mode) {
    case 0: off = tid * 8; break;                  // linear 8B/lane
    case 1: off = (tid & 15) * 8 + (tid >> 4) * 128; break;
    case 2: off = (tid & 15) * 8; break;           // groups share addrs
    case 3:                                        // fragment pattern:
      // rows at stride 144 elems (288 B), group k-base = (tid>>4)*8,
      // want out lane j = col j of rows kb..kb+3 at elem (kb+a)*144+c*4
      off = (((tid >> 4) * 8 + a4) * 144 + c4 * 4) * 2;
      break;
    default: off = tid * 16; break;
  }
  if (addr_mode == 4) {
    // kernel-faithful: 4 fragments x 2 tr reads, one waitcnt at end
    unsigned long long lo[4], hi[4];
    const int kb = (tid >> 4) * 8;
    const unsigned rowb = (unsigned)((kb + a4) * 144 + c4 * 4) * 2;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const unsigned ao = base + rowb + (unsigned)(i * 16) * 2;
      asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(lo[i]) : "v"(ao));
      asm volatile("ds_read_b64_tr_b16 %0, %1" : "=v"(hi[i])
                   : "v"(ao + 4 * 144 * 2));
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    // dump fragment i=2 (the failing rows 32..47) low+high
    unsigned short* o = out + tid * 4;
    o[0] = (unsigned short)(lo[2] & 0xFFFF);
    o[1] = (unsigned short)((lo[2] >> 16) & 0xFFFF);
    o[2] = (unsigned short)(hi[2] & 0xFFFF);
    o[3] = (unsigned short)((hi[2] >> 16) & 0xFFFF);
    return;
  }
  unsigned addr = base + off;
  unsigned long long v;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=v"(v) : "v"(addr));
  unsigned short* o = out + tid * 4;
  o[0] = (unsigned short)(v & 0xFFFF);
  o[1] = (unsigned short)((v >> 16) & 0xFFFF);
  o[2] = (unsigned short)((v >> 32) & 0xFFFF);
  o[3] = (unsigned short)((v >> 48) & 0xFFFF);
}

int main() {
  unsigned short* d;
  (void)hipMalloc(&d, 64 * 4 * 2);
  unsigned short h[256];
  for (int mode = 3; mode < 5; ++mode) {
    hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, d, mode);
    (void)hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    hipError_t e = hipGetLastError();
    printf("== mode %d (%s)\n", mode, hipGetErrorString(e));
    for (int l = 0; l < 64; ++l) {
      printf("lane %2d: %4d %4d %4d %4d\n", l, h[l * 4], h[l * 4 + 1],
             h[l * 4 + 2], h[l * 4 + 3]);
      if ((l & 15) == 15 && l < 63) printf("  --\n");
    }
  }
  return 0;
}
