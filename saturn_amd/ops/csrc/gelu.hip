// Fused tanh-approx GELU forward/backward for CDNA4 (part of SURVEY K1's
// epilogue tail: the round-1 profile showed stock GeluCUDAKernelImpl +
// GeluBackward at ~2.5% of step time on GPT-J;
// profiles/r02_baseline_kernels.txt).
//
// gelu(x)  = 0.5 x (1 + tanh(k (x + c x^3))),  k = sqrt(2/pi), c = 0.044715
// dgelu/dx = 0.5 (1 + t) + 0.5 x (1 - t^2) k (1 + 3 c x^2),  t = tanh(...)
//
// 8 x 16-bit loads/stores per lane (one 16-B dwordx4), grid-stride, tanh
// via the fast exp intrinsic: tanh(u) = 1 - 2 / (exp(2u) + 1).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace samd {

typedef __attribute__((ext_vector_type(8))) short short8v_g;

__device__ __forceinline__ float tanh_fast(float u) {
  return 1.f - 2.f / (__expf(2.f * u) + 1.f);
}

constexpr float GK = 0.7978845608028654f;  // sqrt(2/pi)
constexpr float GC = 0.044715f;

__device__ __forceinline__ float gelu_f(float x) {
  const float t = tanh_fast(GK * (x + GC * x * x * x));
  return 0.5f * x * (1.f + t);
}

__device__ __forceinline__ float dgelu_f(float x) {
  const float u = GK * (x + GC * x * x * x);
  const float t = tanh_fast(u);
  const float du = GK * (1.f + 3.f * GC * x * x);
  return 0.5f * (1.f + t) + 0.5f * x * (1.f - t * t) * du;
}

template <typename T, bool BWD>
__global__ void gelu_kernel(const T* __restrict__ x, const T* __restrict__ dy,
                            T* __restrict__ out, long n) {
  const long nv = n / 8;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nv;
       i += stride) {
    const short8v_g xv = *reinterpret_cast<const short8v_g*>(x + i * 8);
    short8v_g dv;
    if (BWD) dv = *reinterpret_cast<const short8v_g*>(dy + i * 8);
    short8v_g ov;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      T xe;
      xe.x = (unsigned short)xv[e];
      const float xf = toF<T>(xe);
      float r;
      if (BWD) {
        T de;
        de.x = (unsigned short)dv[e];
        r = toF<T>(de) * dgelu_f(xf);
      } else {
        r = gelu_f(xf);
      }
      ov[e] = (short)fromF<T>(r).x;
    }
    *reinterpret_cast<short8v_g*>(out + i * 8) = ov;
  }
  // tail
  const long i0 = blockIdx.x * (long)blockDim.x + threadIdx.x;
  if (i0 == 0) {
    for (long j = nv * 8; j < n; ++j) {
      const float xf = toF<T>(x[j]);
      out[j] = fromF<T>(BWD ? toF<T>(dy[j]) * dgelu_f(xf) : gelu_f(xf));
    }
  }
}

static dim3 grid_1d(long work) {
  const int block = 256;
  return dim3((unsigned)std::max<long>(
      1, std::min<long>((work + block - 1) / block, 8192)));
}

at::Tensor gelu_fwd(at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  auto y = at::empty_like(x);
  const long n = x.numel();
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "gelu_fwd", [&] {
        TORCH_CHECK(sizeof(scalar_t) == 2, "gelu kernel: 16-bit dtypes only");
        if constexpr (sizeof(scalar_t) == 2) {
          hipLaunchKernelGGL((gelu_kernel<scalar_t, false>), grid_1d(n / 8),
                             dim3(256), 0, stream.stream(),
                             reinterpret_cast<const scalar_t*>(x.data_ptr()),
                             (const scalar_t*)nullptr,
                             reinterpret_cast<scalar_t*>(y.data_ptr()), n);
        }
      });
  return y;
}

at::Tensor gelu_bwd(at::Tensor dy, at::Tensor x) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && dy.is_contiguous());
  TORCH_CHECK(dy.sizes() == x.sizes() && dy.scalar_type() == x.scalar_type());
  auto dx = at::empty_like(x);
  const long n = x.numel();
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "gelu_bwd", [&] {
        TORCH_CHECK(sizeof(scalar_t) == 2, "gelu kernel: 16-bit dtypes only");
        if constexpr (sizeof(scalar_t) == 2) {
          hipLaunchKernelGGL((gelu_kernel<scalar_t, true>), grid_1d(n / 8),
                             dim3(256), 0, stream.stream(),
                             reinterpret_cast<const scalar_t*>(x.data_ptr()),
                             reinterpret_cast<const scalar_t*>(dy.data_ptr()),
                             reinterpret_cast<scalar_t*>(dx.data_ptr()), n);
        }
      });
  return dx;
}

}  // namespace samd
