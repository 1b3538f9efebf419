#include "hip/hip_runtime.h"
// Rotary position embedding, fused apply.
//
// Replaces the reference's four-op RoPE chain (GPTJ.py:44-79:
// fixed_pos_embedding einsum + duplicate_interleave + rotate_every_two +
// mul-add) with one elementwise kernel over (b*t, h, pair).  Two layouts:
// interleaved pairs (2i, 2i+1) — GPT-J style — and half-split pairs
// (i, i + rot/2) — Llama/NeoX style.  sin/cos tables are precomputed on
// host (guide App.B: no on-device trig in elementwise kernels) as
// [B*T, rot/2] fp32 rows aligned with x's flattened (B, T).
// Backward is the inverse rotation: same kernel with sign = -1.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace samd {

template <typename T, bool HALF_STYLE>
__global__ void rope_kernel(T* __restrict__ x, const float* __restrict__ cos_t,
                            const float* __restrict__ sin_t, long n_bt,
                            int heads, int dim, int half, float sign) {
  // x: [BT, heads, dim] contiguous, rotated in place on the first 2*half
  // dims of each head vector.
  const long total = n_bt * heads * half;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int p = (int)(i % half);
    const long bth = i / half;
    const int h = (int)(bth % heads);
    const long bt = bth / heads;
    const float c = cos_t[bt * half + p];
    const float s = sin_t[bt * half + p] * sign;
    T* base = x + (bt * heads + h) * (long)dim;
    const int i0 = HALF_STYLE ? p : 2 * p;
    const int i1 = HALF_STYLE ? p + half : 2 * p + 1;
    const float x0 = (float)base[i0];
    const float x1 = (float)base[i1];
    base[i0] = (T)(x0 * c - x1 * s);
    base[i1] = (T)(x1 * c + x0 * s);
  }
}

void rope_apply(at::Tensor x, at::Tensor cos_t, at::Tensor sin_t,
                bool half_style, bool backward) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int dim = (int)x.size(-1);
  const int heads = (int)x.size(-2);
  const long n_bt = x.numel() / ((long)dim * heads);
  const int half = (int)cos_t.size(-1);
  TORCH_CHECK(2 * half <= dim, "rotary dim exceeds head dim");
  TORCH_CHECK(cos_t.numel() == n_bt * half && sin_t.numel() == n_bt * half,
              "cos/sin table rows must match flattened B*T");
  TORCH_CHECK(cos_t.scalar_type() == at::kFloat && cos_t.is_contiguous());
  TORCH_CHECK(sin_t.scalar_type() == at::kFloat && sin_t.is_contiguous());
  auto stream = at::hip::getCurrentHIPStream();
  const long total = n_bt * heads * half;
  const int block = 256;
  dim3 grid((unsigned)std::max<long>(
      1, std::min<long>((total + block - 1) / block, 2048)));
  const float sign = backward ? -1.f : 1.f;
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "rope", [&] {
        auto k = half_style ? rope_kernel<scalar_t, true>
                            : rope_kernel<scalar_t, false>;
        hipLaunchKernelGGL(k, grid, dim3(block), 0, stream.stream(),
                           reinterpret_cast<scalar_t*>(x.data_ptr()),
                           cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
                           n_bt, heads, dim, half, sign);
      });
}

}  // namespace samd
