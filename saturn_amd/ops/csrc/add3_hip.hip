#include "hip/hip_runtime.h"
// Fused 3-way residual add: out = a + b + c in one pass.
// The GPT-J parallel block computes x + attn(h) + mlp(h) (reference
// GPTJ.py:422); stock torch issues two binary adds (3 reads + 2 writes
// extra).  Vectorized 8x16-bit per lane, grid-stride (guide G11/G13).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace samd {

typedef __attribute__((ext_vector_type(8))) short s8v;

template <typename T>
__global__ void add3_kernel(const T* __restrict__ a, const T* __restrict__ b,
                            const T* __restrict__ c, T* __restrict__ out,
                            long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  if constexpr (sizeof(T) == 2) {
    const long nv = n / 8;
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nv;
         i += stride) {
      s8v va = *reinterpret_cast<const s8v*>(a + i * 8);
      s8v vb = *reinterpret_cast<const s8v*>(b + i * 8);
      s8v vc = *reinterpret_cast<const s8v*>(c + i * 8);
      s8v vo;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        T xa, xb, xc;
        xa.x = (unsigned short)va[e];
        xb.x = (unsigned short)vb[e];
        xc.x = (unsigned short)vc[e];
        T r = fromF<T>(toF<T>(xa) + toF<T>(xb) + toF<T>(xc));
        vo[e] = (short)r.x;
      }
      *reinterpret_cast<s8v*>(out + i * 8) = vo;
    }
    for (long i = nv * 8 + blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
         i += stride)
      out[i] = fromF<T>(toF<T>(a[i]) + toF<T>(b[i]) + toF<T>(c[i]));
  } else {
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
         i += stride)
      out[i] = fromF<T>(toF<T>(a[i]) + toF<T>(b[i]) + toF<T>(c[i]));
  }
}

at::Tensor add3(at::Tensor a, at::Tensor b, at::Tensor c) {
  TORCH_CHECK(a.is_cuda() && a.is_contiguous() && b.is_contiguous() &&
              c.is_contiguous());
  TORCH_CHECK(a.sizes() == b.sizes() && a.sizes() == c.sizes());
  auto out = at::empty_like(a);
  const long n = a.numel();
  auto stream = at::hip::getCurrentHIPStream();
  const int block = 256;
  dim3 grid((unsigned)std::max<long>(1, std::min<long>((n / 8 + block - 1) / block, 2048)));
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, a.scalar_type(), "add3", [&] {
        hipLaunchKernelGGL(add3_kernel<scalar_t>, grid, dim3(block), 0,
                           stream.stream(),
                           reinterpret_cast<const scalar_t*>(a.data_ptr()),
                           reinterpret_cast<const scalar_t*>(b.data_ptr()),
                           reinterpret_cast<const scalar_t*>(c.data_ptr()),
                           reinterpret_cast<scalar_t*>(out.data_ptr()), n);
      });
  return out;
}

}  // namespace samd
