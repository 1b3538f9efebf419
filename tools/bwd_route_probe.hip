// Probe: in-register routing of a 16x16 MFMA D-layout tile (S / P^T values)
// into 16x16x32 A-fragments via cross-quarter shfl at lane stride 16 —
// the core exchange of the planned attention-backward v2 (docs/kernels.md
// appendix).  Self-checking PASS/FAIL per (lane, slot), no host math.
//
//   hipcc --offload-arch=gfx950 -O2 tools/bwd_route_probe.hip -o /tmp/brp
//   ./brp     -> "bwd-route PASS" or first mismatch
//
// Layouts (hardware-validated in tools/mfma_probe.hip):
//   D-layout 16x16:  lane l holds D[row=(l>>4)*4+reg][col=l&15], reg 0..3
//   A-layout 16x16x32: lane l supplies A[m=l&15][k=(l>>4)*8+j], j 0..7
//
// Here the D tile is S_n for two 16-q subtiles n=0,1 (rows = q, cols = key);
// the A-fragment wanted is P^T[m=key][k=q over 32 rows].  Target lane l
// (group g=l>>4, key k=l&15) needs the 4-reg payloads of source lanes
//   s0 = a*16+k, s1 = (a+1)*16+k   of subtile n=g>>1, where a=(g&1)*2.

#include <hip/hip_runtime.h>

#include <cstdio>

typedef __attribute__((ext_vector_type(8))) short bf16x8;

__device__ __forceinline__ unsigned short f2us(float f) {
  union {
    float f;
    unsigned u;
  } c{f};
  // round-to-nearest-even bf16
  unsigned r = c.u + 0x7fffu + ((c.u >> 16) & 1);
  return (unsigned short)(r >> 16);
}

__device__ __forceinline__ float us2f(unsigned short u) {
  union {
    unsigned u;
    float f;
  } c;
  c.u = (unsigned)u << 16;
  return c.f;
}

__global__ void probe(int* fails) {
  const int lane = threadIdx.x & 63;
  const int qtr = lane >> 4;   // this lane's quarter as a SOURCE
  const int k = lane & 15;

  // source payloads: subtile n in {0,1}, 4 regs each; encoded exactly in
  // bf16 (|v| <= 255): v = sign(n) * ((qtr*16 + k)*4 + reg)
  float sub[2][4];
#pragma unroll
  for (int n = 0; n < 2; ++n)
#pragma unroll
    for (int reg = 0; reg < 4; ++reg)
      sub[n][reg] = (n ? -1.f : 1.f) * (float)((qtr * 16 + k) * 4 + reg);

  // pack to bf16 pairs (the real kernel ships bf16 to the MFMA anyway)
  unsigned pk[2][2];
#pragma unroll
  for (int n = 0; n < 2; ++n) {
    pk[n][0] = (unsigned)f2us(sub[n][0]) | ((unsigned)f2us(sub[n][1]) << 16);
    pk[n][1] = (unsigned)f2us(sub[n][2]) | ((unsigned)f2us(sub[n][3]) << 16);
  }

  // ---- the routing under test ----
  const int g = lane >> 4;         // target group
  const int a = (g & 1) * 2;       // first source quarter
  const int n = g >> 1;            // source subtile
  const int s0 = a * 16 + k;
  const int s1 = (a + 1) * 16 + k;

  unsigned r[4][2];                // [src(2) x u32(2)] per subtile choice
#pragma unroll
  for (int nn = 0; nn < 2; ++nn) {
    r[nn * 2 + 0][0] = (unsigned)__shfl((int)pk[nn][0], s0, 64);
    r[nn * 2 + 0][1] = (unsigned)__shfl((int)pk[nn][1], s0, 64);
    r[nn * 2 + 1][0] = (unsigned)__shfl((int)pk[nn][0], s1, 64);
    r[nn * 2 + 1][1] = (unsigned)__shfl((int)pk[nn][1], s1, 64);
  }
  bf16x8 frag;
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int src = (j >= 4) ? 1 : 0;          // s0 for slots 0-3, s1 for 4-7
    const int reg = j & 3;
    const unsigned u = r[n * 2 + src][reg >> 1];
    frag[j] = (short)((reg & 1) ? (u >> 16) : (u & 0xffffu));
  }

  // ---- check: frag[j] must be subtile-n value of (quarter a+src, k, reg)
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    const int src = (j >= 4) ? 1 : 0;
    const int reg = j & 3;
    const float expect =
        (n ? -1.f : 1.f) * (float)(((a + src) * 16 + k) * 4 + reg);
    const float got = us2f((unsigned short)frag[j]);
    if (got != expect) {
      atomicAdd(fails, 1);
      if (lane == 0 || atomicAdd(fails, 0) < 4)
        printf("lane %d slot %d: got %f expect %f\n", lane, j, got, expect);
    }
  }
}

int main() {
  int* fails;
  (void)hipMalloc(&fails, sizeof(int));
  (void)hipMemset(fails, 0, sizeof(int));
  hipLaunchKernelGGL(probe, dim3(1), dim3(64), 0, 0, fails);
  int h = -1;
  (void)hipMemcpy(&h, fails, sizeof(int), hipMemcpyDeviceToHost);
  if (h == 0)
    printf("bwd-route PASS\n");
  else
    printf("bwd-route FAIL (%d mismatches)\n", h);
  (void)hipFree(fails);
  return h == 0 ? 0 : 1;
}
