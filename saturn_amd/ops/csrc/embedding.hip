// Token embedding: gather forward + scatter-add backward.
// Reference call sites: GPTJ.py:346,377 (nn.Embedding wte).  The forward is
// a pure HBM row gather (16 B/lane vector copies); the backward accumulates
// per-token rows into an fp32 master table with global atomics (repeated
// tokens collide on the same row, so fp32 atomic accumulation keeps the sum
// exact to fp32 before the single round to the weight dtype), then converts
// once — stock torch instead sorts indices and pays an extra kernel.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace samd {

typedef __attribute__((ext_vector_type(8))) short e8v;

// one block per token row; lanes stride the embedding dim in 8x16-bit chunks
template <typename T>
__global__ void embed_fwd_kernel(const T* __restrict__ w,
                                 const long* __restrict__ idx,
                                 T* __restrict__ out, int E) {
  const long row = idx[blockIdx.x];
  const T* src = w + row * (long)E;
  T* dst = out + blockIdx.x * (long)E;
  if constexpr (sizeof(T) == 2) {
    const int nv = E / 8;
    for (int i = threadIdx.x; i < nv; i += blockDim.x)
      reinterpret_cast<e8v*>(dst)[i] = reinterpret_cast<const e8v*>(src)[i];
    for (int i = nv * 8 + threadIdx.x; i < E; i += blockDim.x) dst[i] = src[i];
  } else {
    for (int i = threadIdx.x; i < E; i += blockDim.x) dst[i] = src[i];
  }
}

template <typename T>
__global__ void embed_bwd_kernel(const T* __restrict__ dout,
                                 const long* __restrict__ idx,
                                 float* __restrict__ acc, int E) {
  const long row = idx[blockIdx.x];
  const T* src = dout + blockIdx.x * (long)E;
  float* dst = acc + row * (long)E;
  if constexpr (sizeof(T) == 2) {
    const int nv = E / 8;
    for (int i = threadIdx.x; i < nv; i += blockDim.x) {
      e8v v = reinterpret_cast<const e8v*>(src)[i];
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        T x;
        x.x = (unsigned short)v[e];
        atomicAdd(dst + i * 8 + e, toF<T>(x));
      }
    }
    for (int i = nv * 8 + threadIdx.x; i < E; i += blockDim.x)
      atomicAdd(dst + i, toF<T>(src[i]));
  } else {
    for (int i = threadIdx.x; i < E; i += blockDim.x)
      atomicAdd(dst + i, toF<T>(src[i]));
  }
}

template <typename T>
__global__ void cast_from_f32_kernel(const float* __restrict__ src,
                                     T* __restrict__ dst, long n) {
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < n;
       i += stride)
    dst[i] = fromF<T>(src[i]);
}

at::Tensor embed_fwd(at::Tensor weight, at::Tensor idx) {
  TORCH_CHECK(weight.is_cuda() && weight.is_contiguous() && weight.dim() == 2);
  TORCH_CHECK(idx.scalar_type() == at::kLong);
  auto idx_flat = idx.contiguous().view(-1);
  const long n = idx_flat.numel();
  const int E = (int)weight.size(1);
  auto sizes = idx.sizes().vec();
  sizes.push_back(E);
  auto out = at::empty(sizes, weight.options());
  if (n == 0) return out;
  auto stream = at::hip::getCurrentHIPStream();
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, weight.scalar_type(), "embed_fwd", [&] {
        hipLaunchKernelGGL(embed_fwd_kernel<scalar_t>, dim3((unsigned)n),
                           dim3(256), 0, stream.stream(),
                           reinterpret_cast<const scalar_t*>(weight.data_ptr()),
                           idx_flat.data_ptr<long>(),
                           reinterpret_cast<scalar_t*>(out.data_ptr()), E);
      });
  return out;
}

at::Tensor embed_bwd(at::Tensor dout, at::Tensor idx, long vocab) {
  TORCH_CHECK(dout.is_cuda());
  auto d = dout.contiguous();
  auto idx_flat = idx.contiguous().view(-1);
  const long n = idx_flat.numel();
  const int E = (int)d.size(-1);
  auto stream = at::hip::getCurrentHIPStream();
  auto acc = at::zeros({vocab, E}, d.options().dtype(at::kFloat));
  if (n > 0) {
    AT_DISPATCH_FLOATING_TYPES_AND2(
        at::kBFloat16, at::kHalf, d.scalar_type(), "embed_bwd", [&] {
          hipLaunchKernelGGL(embed_bwd_kernel<scalar_t>, dim3((unsigned)n),
                             dim3(256), 0, stream.stream(),
                             reinterpret_cast<const scalar_t*>(d.data_ptr()),
                             idx_flat.data_ptr<long>(),
                             acc.data_ptr<float>(), E);
        });
  }
  if (d.scalar_type() == at::kFloat) return acc;
  auto dw = at::empty({vocab, E}, d.options());
  const long total = vocab * (long)E;
  dim3 grid((unsigned)std::max<long>(
      1, std::min<long>((total + 255) / 256, 8192)));
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, d.scalar_type(), "embed_cast", [&] {
        hipLaunchKernelGGL(cast_from_f32_kernel<scalar_t>, grid, dim3(256), 0,
                           stream.stream(), acc.data_ptr<float>(),
                           reinterpret_cast<scalar_t*>(dw.data_ptr()), total);
      });
  return dw;
}

}  // namespace samd
