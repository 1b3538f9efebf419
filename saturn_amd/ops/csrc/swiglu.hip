// Fused SwiGLU elementwise: out = silu(gate) * up, plus the backward pair
// (dgate = g * up * dsilu(gate), dup = g * silu(gate)).  Stock torch issues
// separate silu and mul kernels each way (extra HBM passes on the 14336-
// wide Llama FFN activations).  Vectorized 8x16-bit IO per lane (G13).

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace samd {

typedef __attribute__((ext_vector_type(8))) short s8w;

__device__ __forceinline__ float silu_f(float x) {
  return x / (1.f + __expf(-x));
}

template <typename T>
__global__ void swiglu_fwd_kernel(const T* __restrict__ gate,
                                  const T* __restrict__ up,
                                  T* __restrict__ out, long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  if constexpr (sizeof(T) == 2) {
    const long nv = n / 8;
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nv;
         i += stride) {
      s8w vg = *reinterpret_cast<const s8w*>(gate + i * 8);
      s8w vu = *reinterpret_cast<const s8w*>(up + i * 8);
      s8w vo;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float g = us2f((unsigned short)vg[e]);
        const float u = us2f((unsigned short)vu[e]);
        vo[e] = (short)f2us(silu_f(g) * u);
      }
      *reinterpret_cast<s8w*>(out + i * 8) = vo;
    }
    for (long i = nv * 8 + blockIdx.x * (long)blockDim.x + threadIdx.x;
         i < n; i += stride)
      out[i] = fromF<T>(silu_f(toF<T>(gate[i])) * toF<T>(up[i]));
  } else {
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
         i += stride)
      out[i] = fromF<T>(silu_f(toF<T>(gate[i])) * toF<T>(up[i]));
  }
}

template <typename T>
__global__ void swiglu_bwd_kernel(const T* __restrict__ dout,
                                  const T* __restrict__ gate,
                                  const T* __restrict__ up,
                                  T* __restrict__ dgate,
                                  T* __restrict__ dup, long n) {
  // vectorized like the forward: the round-1 backward used scalar 2-byte
  // loads (G13) — 233 us vs a ~150 us traffic floor at the Llama FFN shape
  const long stride = (long)gridDim.x * blockDim.x;
  if constexpr (sizeof(T) == 2) {
    const long nv = n / 8;
    for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < nv;
         i += stride) {
      s8w vg = *reinterpret_cast<const s8w*>(gate + i * 8);
      s8w vu = *reinterpret_cast<const s8w*>(up + i * 8);
      s8w vo = *reinterpret_cast<const s8w*>(dout + i * 8);
      s8w odg, odu;
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const float g = us2f((unsigned short)vg[e]);
        const float u = us2f((unsigned short)vu[e]);
        const float go = us2f((unsigned short)vo[e]);
        const float sig = 1.f / (1.f + __expf(-g));
        // d silu(g)/dg = sig * (1 + g * (1 - sig))
        odg[e] = (short)f2us(go * u * sig * (1.f + g * (1.f - sig)));
        odu[e] = (short)f2us(go * g * sig);
      }
      *reinterpret_cast<s8w*>(dgate + i * 8) = odg;
      *reinterpret_cast<s8w*>(dup + i * 8) = odu;
    }
    for (long i = nv * 8 + blockIdx.x * (long)blockDim.x + threadIdx.x;
         i < n; i += stride) {
      const float g = toF<T>(gate[i]);
      const float u = toF<T>(up[i]);
      const float go = toF<T>(dout[i]);
      const float sig = 1.f / (1.f + __expf(-g));
      dgate[i] = fromF<T>(go * u * sig * (1.f + g * (1.f - sig)));
      dup[i] = fromF<T>(go * g * sig);
    }
    return;
  }
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride) {
    const float g = toF<T>(gate[i]);
    const float u = toF<T>(up[i]);
    const float go = toF<T>(dout[i]);
    const float sig = 1.f / (1.f + __expf(-g));
    const float si = g * sig;
    // d silu(g)/dg = sig * (1 + g * (1 - sig))
    dgate[i] = fromF<T>(go * u * sig * (1.f + g * (1.f - sig)));
    dup[i] = fromF<T>(go * si);
  }
}

at::Tensor swiglu_fwd(at::Tensor gate, at::Tensor up) {
  TORCH_CHECK(gate.is_cuda() && gate.is_contiguous() && up.is_contiguous());
  TORCH_CHECK(gate.sizes() == up.sizes());
  auto out = at::empty_like(gate);
  const long n = gate.numel();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid((unsigned)std::max<long>(
      1, std::min<long>((n / 8 + 255) / 256, 2048))), block(256);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gate.scalar_type(), "swiglu_fwd", [&] {
        hipLaunchKernelGGL(swiglu_fwd_kernel<scalar_t>, grid, block, 0,
                           stream.stream(),
                           reinterpret_cast<const scalar_t*>(gate.data_ptr()),
                           reinterpret_cast<const scalar_t*>(up.data_ptr()),
                           reinterpret_cast<scalar_t*>(out.data_ptr()), n);
      });
  return out;
}

std::vector<at::Tensor> swiglu_bwd(at::Tensor dout, at::Tensor gate,
                                   at::Tensor up) {
  auto dgate = at::empty_like(gate);
  auto dup = at::empty_like(up);
  const long n = gate.numel();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid((unsigned)std::max<long>(
      1, std::min<long>((n / 8 + 255) / 256, 4096))), block(256);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, gate.scalar_type(), "swiglu_bwd", [&] {
        hipLaunchKernelGGL(swiglu_bwd_kernel<scalar_t>, grid, block, 0,
                           stream.stream(),
                           reinterpret_cast<const scalar_t*>(dout.data_ptr()),
                           reinterpret_cast<const scalar_t*>(gate.data_ptr()),
                           reinterpret_cast<const scalar_t*>(up.data_ptr()),
                           reinterpret_cast<scalar_t*>(dgate.data_ptr()),
                           reinterpret_cast<scalar_t*>(dup.data_ptr()), n);
      });
  return {dgate, dup};
}

}  // namespace samd
