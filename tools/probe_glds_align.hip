// Probe: does __builtin_amdgcn_global_load_lds(.., 16, ..) accept an
// 8-byte-aligned (but not 16-byte-aligned) GLOBAL source on gfx950?
// The C=4 im2col staging produces such sources (pixel stride = 8 B).
#include <hip/hip_runtime.h>
#include <cstdio>

__global__ void probe_k(const unsigned short* src, unsigned short* out,
                        int shift) {
  __shared__ unsigned short lds[64 * 8];
  const int lane = threadIdx.x & 63;
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) void*)(src + shift +
                                                      lane * 8),
      (__attribute__((address_space(3))) void*)(lds + lane * 8), 16, 0,
      0);
  asm volatile("s_waitcnt vmcnt(0)" ::: "memory");
  __syncthreads();
  for (int j = 0; j < 8; ++j) out[lane * 8 + j] = lds[lane * 8 + j];
}

int main() {
  unsigned short *src, *out, h[1024];
  (void)hipMalloc(&src, 4096);
  (void)hipMalloc(&out, 4096);
  for (int i = 0; i < 1024; ++i) h[i] = (unsigned short)(i * 3 + 7);
  (void)hipMemcpy(src, h, 2048, hipMemcpyHostToDevice);
  int bad_total = 0;
  for (int shift : {0, 4, 2}) {   // 0B, 8B, 4B misalignment (elements)
    hipLaunchKernelGGL(probe_k, dim3(1), dim3(64), 0, 0, src, out,
                       shift);
    (void)hipDeviceSynchronize();
    unsigned short r[512];
    (void)hipMemcpy(r, out, 1024, hipMemcpyDeviceToHost);
    int bad = 0;
    for (int i = 0; i < 512; ++i)
      if (r[i] != h[shift + i]) ++bad;
    printf("shift %d elems (%d B): %s (%d bad)\n", shift, shift * 2,
           bad ? "FAIL" : "OK", bad);
    bad_total += bad;
  }
  return bad_total ? 1 : 0;
}
