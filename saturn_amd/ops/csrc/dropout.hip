// Fused dropout (reference GPTJ.py:95-96,347 attn/resid/embd
// dropout).  Counter-based RNG keyed on (seed, element index): the backward
// regenerates the exact keep mask from the saved 64-bit seed, so no mask
// tensor is ever written — both passes are single-read/single-write
// HBM-bound kernels.  The generator is a splitmix64 finalizer (full
// avalanche; each element's keep decision is an independent uniform draw),
// which is ample for dropout and costs ~10 VALU ops per element group.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace samd {

typedef __attribute__((ext_vector_type(8))) short d8v;

__device__ __forceinline__ unsigned long long mix64(unsigned long long z) {
  z += 0x9e3779b97f4a7c15ull;
  z = (z ^ (z >> 30)) * 0xbf58476d1ce4e5b9ull;
  z = (z ^ (z >> 27)) * 0x94d049bb133111ebull;
  return z ^ (z >> 31);
}

// two elements per 64-bit draw: lane handles 8 elements = 4 draws
template <typename T, bool BWD>
__global__ void dropout_kernel(const T* __restrict__ x, T* __restrict__ out,
                               long n, unsigned long long seed,
                               unsigned threshold, float scale) {
  const long stride = (long)gridDim.x * blockDim.x;
  const long nv = n / 8;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nv;
       i += stride) {
    d8v v = *reinterpret_cast<const d8v*>(x + i * 8);
    d8v o;
#pragma unroll
    for (int d = 0; d < 4; ++d) {
      const unsigned long long r = mix64(seed ^ (unsigned long long)(i * 4 + d));
      const unsigned lo = (unsigned)r, hi = (unsigned)(r >> 32);
#pragma unroll
      for (int e = 0; e < 2; ++e) {
        const unsigned u = e ? hi : lo;
        T xv;
        xv.x = (unsigned short)v[d * 2 + e];
        T r2 = fromF<T>(u >= threshold ? toF<T>(xv) * scale : 0.f);
        o[d * 2 + e] = (short)r2.x;
      }
    }
    *reinterpret_cast<d8v*>(out + i * 8) = o;
  }
  // tail: same indexing contract (element j belongs to draw j/2)
  for (long j = nv * 8 + blockIdx.x * (long)blockDim.x + threadIdx.x; j < n;
       j += stride) {
    const unsigned long long r = mix64(seed ^ (unsigned long long)(j / 2));
    const unsigned u = (j & 1) ? (unsigned)(r >> 32) : (unsigned)r;
    out[j] = fromF<T>(u >= threshold ? toF<T>(x[j]) * scale : 0.f);
  }
}

static at::Tensor dropout_apply(at::Tensor x, double p, long seed) {
  TORCH_CHECK(x.is_cuda() && p >= 0.0 && p < 1.0);
  auto xc = x.contiguous();
  auto out = at::empty_like(xc);
  const long n = xc.numel();
  if (n == 0) return out;
  const unsigned threshold = (unsigned)(p * 4294967296.0);
  const float scale = (float)(1.0 / (1.0 - p));
  auto stream = at::hip::getCurrentHIPStream();
  const int block = 256;
  dim3 grid((unsigned)std::max<long>(
      1, std::min<long>((n / 8 + block - 1) / block, 4096)));
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, xc.scalar_type(), "dropout", [&] {
        if constexpr (sizeof(scalar_t) == 2) {
          hipLaunchKernelGGL((dropout_kernel<scalar_t, false>), grid,
                             dim3(block), 0, stream.stream(),
                             reinterpret_cast<const scalar_t*>(xc.data_ptr()),
                             reinterpret_cast<scalar_t*>(out.data_ptr()), n,
                             (unsigned long long)seed, threshold, scale);
        } else {
          TORCH_CHECK(false, "dropout kernel: 16-bit dtypes only (bf16/fp16)");
        }
      });
  return out;
}

at::Tensor dropout_fwd(at::Tensor x, double p, long seed) {
  return dropout_apply(x, p, seed);
}

// backward = the same masked scale applied to the incoming grad
at::Tensor dropout_bwd(at::Tensor dout, double p, long seed) {
  return dropout_apply(dout, p, seed);
}

}  // namespace samd
