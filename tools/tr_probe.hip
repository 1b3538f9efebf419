// Probe ds_read_b64_tr_b16 gather semantics on gfx950.
//
// Guide formula: with per-lane address addr(l), element j of lane l comes
// from lds bytes addr-relative — canonical pattern: per-lane addr =
// base + [(l&15) + (l>>4)*64] * 2 and element j reads element index
// (l&15) + j*16 + (l>>4)*64 of the image (16-bit units, stride 32 B
// between elements).  This probe fills LDS with idx values and prints the
// observed (lane, j) -> element-index map for two address hypotheses.
//
//   hipcc --offload-arch=gfx950 -O2 tools/tr_probe.hip -o /tmp/tr_probe

#include <hip/hip_runtime.h>
#include <cstdio>

typedef __attribute__((ext_vector_type(4))) short short4v;

__global__ void k(short* out, int hyp) {
  __shared__ unsigned short img[1024];
  for (int i = threadIdx.x; i < 1024; i += blockDim.x) img[i] = (unsigned short)i;
  __syncthreads();
  const int l = threadIdx.x;
  unsigned off;
  if (hyp == 0) off = ((l & 15) + (l >> 4) * 64) * 2;      // linear in lane
  else if (hyp == 1) off = (((l & 15) * 3) % 16 + (l >> 4) * 64) * 2;  // permuted
  else if (hyp == 2) off = 128 * 2 * (l >> 4);             // uniform per group
  else off = ((l & 15) * 16 + (l >> 4) * 512) * 2 % 2048;  // stride-16 rows
  // ds ops take a 32-bit LDS byte address: addrspacecast to AS(3), truncate
  unsigned a32 =
      (unsigned)(unsigned long long)(__attribute__((address_space(3)))
                                         unsigned short*)&img[0] +
      off;
  short4v v;
  asm volatile("ds_read_b64_tr_b16 %0, %1\n\ts_waitcnt lgkmcnt(0)"
               : "=&v"(v)
               : "v"(a32)
               : "memory");
  for (int j = 0; j < 4; ++j) out[l * 4 + j] = v[j];
}

int main() {
  short* d;
  hipMalloc(&d, 64 * 4 * 2);
  for (int hyp = 0; hyp < 4; ++hyp) {
    hipMemset(d, 0xff, 64 * 4 * 2);
    hipLaunchKernelGGL(k, dim3(1), dim3(64), 0, 0, d, hyp);
    short h[256];
    hipMemcpy(h, d, sizeof(h), hipMemcpyDeviceToHost);
    printf("hyp %d:\n", hyp);
    for (int l = 0; l < 32; l += 1) {
      printf("  l%02d:", l);
      for (int j = 0; j < 4; ++j) printf(" %4d", h[l * 4 + j]);
      printf("\n");
    }
  }
  return 0;
}
