// MFMA fragment-layout probe for gfx950.
//
// Validates the lane->element mappings this repo's MFMA kernels assume for
// v_mfma_f32_16x16x32_bf16 and v_mfma_f32_32x32x16_bf16 by computing a full
// tile against a host reference with ASYMMETRIC inputs (guide §3: symmetric
// inputs miss transposes).  Prints PASS/FAIL per (A-hyp, B-hyp) pair.
//
//   hipcc --offload-arch=gfx950 -O2 tools/mfma_probe.hip -o /tmp/mfma_probe
//   /tmp/mfma_probe

#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <cstdio>
#include <cmath>
#include <vector>

typedef __attribute__((ext_vector_type(8))) short bf16x8;
typedef __attribute__((ext_vector_type(4))) float f32x4;
typedef __attribute__((ext_vector_type(16))) float f32x16;

static unsigned short f2us(float f) {
  union { unsigned int i; float f; } c; c.f = f;
  unsigned int lsb = (c.i >> 16) & 1u;
  return (unsigned short)((c.i + 0x7fffu + lsb) >> 16);
}
static float us2f(unsigned short u) {
  union { unsigned int i; float f; } c; c.i = ((unsigned int)u) << 16;
  return c.f;
}

// ---- 16x16x32: A[16][32], B[32][16], D[16][16] ----------------------------
// A hyp h: element index for lane l, slot j (j=0..7):
//   h=0: A[l&15][(l>>4)*8 + j]
//   h=1: A[l&15][j*4 + (l>>4)]
//   h=2: A[l&15][(l>>4) + 4*j]   (same as h=1? no: j*4+(l>>4) == (l>>4)+4*j; keep 2 variants below)
//   h=2: A[l&15][2*((l>>4)*4 + (j>>1)) + (j&1)]  (paired k)
__global__ void k16(const unsigned short* A, const unsigned short* B, float* D,
                    int ah, int bh) {
  int l = threadIdx.x;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    int r = l & 15, q = l >> 4;
    int ka = (ah == 0) ? (q * 8 + j) : (ah == 1) ? (j * 4 + q)
                       : (2 * (q * 4 + (j >> 1)) + (j & 1));
    int kb = (bh == 0) ? (q * 8 + j) : (bh == 1) ? (j * 4 + q)
                       : (2 * (q * 4 + (j >> 1)) + (j & 1));
    a[j] = (short)A[r * 32 + ka];
    b[j] = (short)B[kb * 16 + r];
  }
  f32x4 d = {0.f, 0.f, 0.f, 0.f};
  d = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, d, 0, 0, 0);
  for (int reg = 0; reg < 4; ++reg) {
    int col = l & 15, row = (l >> 4) * 4 + reg;
    D[row * 16 + col] = d[reg];
  }
}

// ---- 32x32x16: A[32][16], B[16][32], D[32][32] ----------------------------
__global__ void k32(const unsigned short* A, const unsigned short* B, float* D,
                    int ah, int bh) {
  int l = threadIdx.x;
  bf16x8 a, b;
  for (int j = 0; j < 8; ++j) {
    int r = l & 31, q = l >> 5;  // q in {0,1}
    int ka = (ah == 0) ? (q * 8 + j) : (ah == 1) ? (j * 2 + q)
                       : (2 * (q * 4 + (j >> 1)) + (j & 1));
    int kb = (bh == 0) ? (q * 8 + j) : (bh == 1) ? (j * 2 + q)
                       : (2 * (q * 4 + (j >> 1)) + (j & 1));
    a[j] = (short)A[r * 16 + ka];
    b[j] = (short)B[kb * 32 + r];
  }
  f32x16 d;
  for (int i = 0; i < 16; ++i) d[i] = 0.f;
  d = __builtin_amdgcn_mfma_f32_32x32x16_bf16(a, b, d, 0, 0, 0);
  for (int reg = 0; reg < 16; ++reg) {
    int col = l & 31;
    int row = (reg & 3) + 8 * (reg >> 2) + 4 * (l >> 5);
    D[row * 32 + col] = d[reg];
  }
}

static bool check(const std::vector<float>& D, const std::vector<float>& ref,
                  int n) {
  for (int i = 0; i < n; ++i)
    if (fabsf(D[i] - ref[i]) > 1e-2f * (1.f + fabsf(ref[i]))) return false;
  return true;
}

int main() {
  // 16x16x32
  {
    int M = 16, N = 16, K = 32;
    std::vector<unsigned short> A(M * K), B(K * N);
    std::vector<float> Af(M * K), Bf(K * N), ref(M * N);
    for (int i = 0; i < M * K; ++i) { Af[i] = 0.01f * i - 1.3f; A[i] = f2us(Af[i]); }
    for (int i = 0; i < K * N; ++i) { Bf[i] = 0.02f * i - 3.7f; B[i] = f2us(Bf[i]); }
    for (int r = 0; r < M; ++r)
      for (int c = 0; c < N; ++c) {
        float s = 0;
        for (int k = 0; k < K; ++k)
          s += us2f(f2us(Af[r * K + k])) * us2f(f2us(Bf[k * N + c]));
        ref[r * N + c] = s;
      }
    unsigned short *dA, *dB; float* dD;
    hipMalloc(&dA, A.size() * 2); hipMalloc(&dB, B.size() * 2);
    hipMalloc(&dD, M * N * 4);
    hipMemcpy(dA, A.data(), A.size() * 2, hipMemcpyHostToDevice);
    hipMemcpy(dB, B.data(), B.size() * 2, hipMemcpyHostToDevice);
    for (int ah = 0; ah < 3; ++ah)
      for (int bh = 0; bh < 3; ++bh) {
        hipMemset(dD, 0, M * N * 4);
        hipLaunchKernelGGL(k16, dim3(1), dim3(64), 0, 0, dA, dB, dD, ah, bh);
        std::vector<float> D(M * N);
        hipMemcpy(D.data(), dD, M * N * 4, hipMemcpyDeviceToHost);
        printf("16x16x32 A-hyp %d B-hyp %d: %s\n", ah, bh,
               check(D, ref, M * N) ? "PASS" : "fail");
      }
    hipFree(dA); hipFree(dB); hipFree(dD);
  }
  // 32x32x16
  {
    int M = 32, N = 32, K = 16;
    std::vector<unsigned short> A(M * K), B(K * N);
    std::vector<float> Af(M * K), Bf(K * N), ref(M * N);
    for (int i = 0; i < M * K; ++i) { Af[i] = 0.013f * i - 2.1f; A[i] = f2us(Af[i]); }
    for (int i = 0; i < K * N; ++i) { Bf[i] = 0.017f * i - 1.9f; B[i] = f2us(Bf[i]); }
    for (int r = 0; r < M; ++r)
      for (int c = 0; c < N; ++c) {
        float s = 0;
        for (int k = 0; k < K; ++k)
          s += us2f(f2us(Af[r * K + k])) * us2f(f2us(Bf[k * N + c]));
        ref[r * N + c] = s;
      }
    unsigned short *dA, *dB; float* dD;
    hipMalloc(&dA, A.size() * 2); hipMalloc(&dB, B.size() * 2);
    hipMalloc(&dD, M * N * 4);
    hipMemcpy(dA, A.data(), A.size() * 2, hipMemcpyHostToDevice);
    hipMemcpy(dB, B.data(), B.size() * 2, hipMemcpyHostToDevice);
    for (int ah = 0; ah < 3; ++ah)
      for (int bh = 0; bh < 3; ++bh) {
        hipMemset(dD, 0, M * N * 4);
        hipLaunchKernelGGL(k32, dim3(1), dim3(64), 0, 0, dA, dB, dD, ah, bh);
        std::vector<float> D(M * N);
        hipMemcpy(D.data(), dD, M * N * 4, hipMemcpyDeviceToHost);
        printf("32x32x16 A-hyp %d B-hyp %d: %s\n", ah, bh,
               check(D, ref, M * N) ? "PASS" : "fail");
      }
    hipFree(dA); hipFree(dB); hipFree(dD);
  }
  return 0;
}
