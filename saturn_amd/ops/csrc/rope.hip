// Rotary position embedding, fused apply.
//
// Replaces the reference's four-op RoPE chain (GPTJ.py:44-79:
// fixed_pos_embedding einsum + duplicate_interleave + rotate_every_two +
// mul-add) with one out-of-place elementwise kernel.  Two layouts:
// interleaved pairs (2i, 2i+1) — GPT-J style — and half-split pairs
// (i, i + rot/2) — Llama/NeoX style.
//
// Round-2 redesign: the round-1 kernel was in-place over a defensive
// .clone() with a pre-materialized [B,T,half] table — 2x the HBM traffic
// (the clone reads+writes the whole tensor, then the kernel touches the
// rotary slice again) plus an expand+contiguous per call.  This version
// writes y directly from x in 16-B groups (rotary part rotated, tail
// copied) and reads the shared fp32 [T_total, half] table with a position
// offset (the offset is how sequence-parallel shards index their global
// positions).  Backward is the inverse rotation: same kernel, sign = -1.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace samd {

typedef __attribute__((ext_vector_type(8))) short short8v_r;

// Vector path: dim % 8 == 0 and the rotary region splits on 16-B group
// boundaries.  One lane handles 8 contiguous elements of one head vector.
template <typename T, bool HALF_STYLE>
__global__ void rope_oop_vec_kernel(const T* __restrict__ x, T* __restrict__ y,
                                    const float* __restrict__ cos_t,
                                    const float* __restrict__ sin_t, long n_bt,
                                    int t_len, long t_off, int heads, int dim,
                                    int half, float sign) {
  const int gdim = dim / 8;
  const long total = n_bt * heads * gdim;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int gj = (int)(i % gdim);
    const long bth = i / gdim;
    const int h = (int)(bth % heads);
    const long bt = bth / heads;
    const long base = (bt * heads + h) * (long)dim;
    const int j0 = gj * 8;
    const short8v_r xv = *reinterpret_cast<const short8v_r*>(x + base + j0);
    if (j0 >= 2 * half) {  // non-rotary tail: straight 16-B copy
      *reinterpret_cast<short8v_r*>(y + base + j0) = xv;
      continue;
    }
    const long row = (t_off + (bt % t_len)) * (long)half;
    short8v_r ov;
    if constexpr (!HALF_STYLE) {
      // pairs (2p, 2p+1) live inside this vector
#pragma unroll
      for (int e = 0; e < 4; ++e) {
        const int p = j0 / 2 + e;
        const float c = cos_t[row + p];
        const float s = sin_t[row + p] * sign;
        T e0, e1;
        e0.x = (unsigned short)xv[2 * e];
        e1.x = (unsigned short)xv[2 * e + 1];
        const float x0 = toF<T>(e0);
        const float x1 = toF<T>(e1);
        ov[2 * e] = (short)fromF<T>(x0 * c - x1 * s).x;
        ov[2 * e + 1] = (short)fromF<T>(x1 * c + x0 * s).x;
      }
    } else {
      // partner vector lives at j0 +/- half (half % 8 == 0)
      const bool low = j0 < half;
      const short8v_r pv = *reinterpret_cast<const short8v_r*>(
          x + base + (low ? j0 + half : j0 - half));
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        const int j = j0 + e;
        const int p = low ? j : j - half;
        const float c = cos_t[row + p];
        const float s = sin_t[row + p] * sign;
        T own, par;
        own.x = (unsigned short)xv[e];
        par.x = (unsigned short)pv[e];
        const float xo = toF<T>(own);
        const float xp = toF<T>(par);
        // low half: y = x0*c - x1*s;  high half: y = x1*c + x0*s
        ov[e] = (short)fromF<T>(low ? xo * c - xp * s : xo * c + xp * s).x;
      }
    }
    *reinterpret_cast<short8v_r*>(y + base + j0) = ov;
  }
}

// Scalar path: any dim / half (tiny test models).
template <typename T, bool HALF_STYLE>
__global__ void rope_oop_kernel(const T* __restrict__ x, T* __restrict__ y,
                                const float* __restrict__ cos_t,
                                const float* __restrict__ sin_t, long n_bt,
                                int t_len, long t_off, int heads, int dim,
                                int half, float sign) {
  const long total = n_bt * heads * dim;
  const long stride = (long)gridDim.x * blockDim.x;
  for (long i = blockIdx.x * (long)blockDim.x + threadIdx.x; i < total;
       i += stride) {
    const int j = (int)(i % dim);
    const long bth = i / dim;
    const int h = (int)(bth % heads);
    const long bt = bth / heads;
    const long base = (bt * heads + h) * (long)dim;
    if (j >= 2 * half) {
      y[base + j] = x[base + j];
      continue;
    }
    const long row = (t_off + (bt % t_len)) * (long)half;
    int p, jp;
    bool low;
    if (HALF_STYLE) {
      low = j < half;
      p = low ? j : j - half;
      jp = low ? j + half : j - half;
    } else {
      low = (j % 2) == 0;
      p = j / 2;
      jp = low ? j + 1 : j - 1;
    }
    const float c = cos_t[row + p];
    const float s = sin_t[row + p] * sign;
    const float xo = toF<T>(x[base + j]);
    const float xp = toF<T>(x[base + jp]);
    y[base + j] = fromF<T>(low ? xo * c - xp * s : xo * c + xp * s);
  }
}

void rope_apply(at::Tensor y, at::Tensor x, at::Tensor cos_t, at::Tensor sin_t,
                int64_t t_len, int64_t t_off, bool half_style, bool backward) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous() && y.is_contiguous());
  TORCH_CHECK(y.sizes() == x.sizes() && y.scalar_type() == x.scalar_type());
  const int dim = (int)x.size(-1);
  const int heads = (int)x.size(-2);
  const long n_bt = x.numel() / ((long)dim * heads);
  const int half = (int)cos_t.size(-1);
  TORCH_CHECK(2 * half <= dim, "rotary dim exceeds head dim");
  TORCH_CHECK(n_bt % t_len == 0, "flattened B*T not a multiple of T");
  TORCH_CHECK(cos_t.size(0) >= t_off + t_len && sin_t.size(0) >= t_off + t_len,
              "rope table too short for offset+T");
  TORCH_CHECK(cos_t.scalar_type() == at::kFloat && cos_t.is_contiguous());
  TORCH_CHECK(sin_t.scalar_type() == at::kFloat && sin_t.is_contiguous());
  auto stream = at::hip::getCurrentHIPStream();
  const float sign = backward ? -1.f : 1.f;
  const bool vec_ok = x.scalar_type() != at::kFloat && dim % 8 == 0 &&
                      (half_style ? half % 8 == 0 : (2 * half) % 8 == 0);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "rope", [&] {
        const long total =
            vec_ok ? n_bt * heads * (dim / 8) : n_bt * heads * (long)dim;
        const int block = 256;
        dim3 grid((unsigned)std::max<long>(
            1, std::min<long>((total + block - 1) / block, 4096)));
        if (vec_ok) {
          if constexpr (sizeof(scalar_t) == 2) {
            auto k = half_style ? rope_oop_vec_kernel<scalar_t, true>
                                : rope_oop_vec_kernel<scalar_t, false>;
            hipLaunchKernelGGL(
                k, grid, dim3(block), 0, stream.stream(),
                reinterpret_cast<const scalar_t*>(x.data_ptr()),
                reinterpret_cast<scalar_t*>(y.data_ptr()),
                cos_t.data_ptr<float>(), sin_t.data_ptr<float>(), n_bt,
                (int)t_len, (long)t_off, heads, dim, half, sign);
            return;
          }
        }
        auto k = half_style ? rope_oop_kernel<scalar_t, true>
                            : rope_oop_kernel<scalar_t, false>;
        hipLaunchKernelGGL(k, grid, dim3(block), 0, stream.stream(),
                           reinterpret_cast<const scalar_t*>(x.data_ptr()),
                           reinterpret_cast<scalar_t*>(y.data_ptr()),
                           cos_t.data_ptr<float>(), sin_t.data_ptr<float>(),
                           n_bt, (int)t_len, (long)t_off, heads, dim, half,
                           sign);
      });
}

}  // namespace samd
