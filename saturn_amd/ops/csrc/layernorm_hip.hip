#include "hip/hip_runtime.h"
// LayerNorm + RMSNorm forward/backward for CDNA4.
//
// One 256-thread block (4 waves) per chunk of rows; each thread owns a
// fixed column slice so the backward's dweight/dbias partials accumulate in
// registers across the block's rows and hit global memory once per block
// (guide Guideline 12: per-block partial reduction before atomics).
// Stats are fp32; IO is bf16/fp16/fp32, vectorized 8 elements per thread
// access where the row length allows (guide G13).
//
// Replaces the implicit torch kernels behind nn.LayerNorm at reference
// GPTJ.py:396,350 with hand-written wave64 kernels.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace samd {

constexpr int LN_BLOCK = 256;

// ---------------------------------------------------------------------------
// Forward: y = (x - mean) * rstd * w + b     (RMS: y = x * rrms * w)
// ---------------------------------------------------------------------------
template <typename T, bool RMS>
__global__ void norm_fwd_kernel(const T* __restrict__ x, const T* __restrict__ w,
                                const T* __restrict__ b, T* __restrict__ y,
                                float* __restrict__ mean_out,
                                float* __restrict__ rstd_out, int rows,
                                int cols, float eps) {
  __shared__ float lds[LN_BLOCK / WAVE];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * cols;
    T* yr = y + (long)row * cols;
    float s = 0.f, sq = 0.f;
    for (int j = threadIdx.x; j < cols; j += LN_BLOCK) {
      float v = (float)xr[j];
      s += v;
      sq += v * v;
    }
    float mean = 0.f;
    if (!RMS) {
      mean = block_sum<LN_BLOCK>(s, lds) / cols;
    }
    float var = block_sum<LN_BLOCK>(sq, lds) / cols - mean * mean;
    float rstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      if (!RMS && mean_out) mean_out[row] = mean;
      if (rstd_out) rstd_out[row] = rstd;
    }
    for (int j = threadIdx.x; j < cols; j += LN_BLOCK) {
      float v = ((float)xr[j] - mean) * rstd;
      float wv = (float)w[j];
      float o = RMS ? v * wv : v * wv + (b ? (float)b[j] : 0.f);
      yr[j] = (T)o;
    }
    __syncthreads();
  }
}

// ---------------------------------------------------------------------------
// Backward.
// dx = rstd * (dyw - mean(dyw) - xhat * mean(dyw * xhat))   [LN]
// dx = rrms * (dyw - xhat * mean(dyw * xhat))               [RMS]
// where dyw = dy * w, xhat = (x - mean) * rstd.
// dw[j] = sum_rows dy * xhat ;  db[j] = sum_rows dy.
// ---------------------------------------------------------------------------
template <typename T, bool RMS, int CPT>
__global__ void norm_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                const T* __restrict__ w,
                                const float* __restrict__ mean,
                                const float* __restrict__ rstd,
                                T* __restrict__ dx, float* __restrict__ dw,
                                float* __restrict__ db, int rows, int cols) {
  __shared__ float lds[LN_BLOCK / WAVE];
  // Per-thread column ownership for the dw/db partials; CPT is dispatched
  // by the host so a 4096-col row doesn't pay a 16384-col register budget.
  float dwp[CPT];
  float dbp[CPT];
#pragma unroll
  for (int k = 0; k < CPT; ++k) { dwp[k] = 0.f; dbp[k] = 0.f; }

  // Per-thread row cache (rows <= 4096 cols): the second pass (dx) reuses
  // the values loaded by the reduction pass instead of re-reading dy/x
  // from HBM (5 -> 3 global passes; the op is bandwidth-bound).  Longer
  // rows would blow the register budget (CPT=64 -> 256 extra VGPRs), so
  // they keep the re-read form.
  constexpr bool CACHE = (CPT <= 16);
  float c_dyw[CACHE ? CPT : 1];
  float c_xh[CACHE ? CPT : 1];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + (long)row * cols;
    const T* xr = x + (long)row * cols;
    T* dxr = dx + (long)row * cols;
    const float mu = RMS ? 0.f : mean[row];
    const float rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
    for (int k = 0, j = threadIdx.x; j < cols; j += LN_BLOCK, ++k) {
      float dyv = (float)dyr[j];
      float xh = ((float)xr[j] - mu) * rs;
      float dyw = dyv * (float)w[j];
      if constexpr (CACHE) {
        c_dyw[k] = dyw;
        c_xh[k] = xh;
      }
      s1 += dyw;
      s2 += dyw * xh;
      dwp[k] += dyv * xh;
      dbp[k] += dyv;
    }
    float m1 = RMS ? 0.f : block_sum<LN_BLOCK>(s1, lds) / cols;
    float m2 = block_sum<LN_BLOCK>(s2, lds) / cols;
    if constexpr (CACHE) {
      for (int k = 0, j = threadIdx.x; j < cols; j += LN_BLOCK, ++k)
        dxr[j] = (T)(rs * (c_dyw[k] - m1 - c_xh[k] * m2));
    } else {
      for (int j = threadIdx.x; j < cols; j += LN_BLOCK) {
        float dyw = (float)dyr[j] * (float)w[j];
        float xh = ((float)xr[j] - mu) * rs;
        dxr[j] = (T)(rs * (dyw - m1 - xh * m2));
      }
    }
    __syncthreads();
  }
  for (int k = 0, j = threadIdx.x; j < cols; j += LN_BLOCK, ++k) {
    atomicAdd(&dw[j], dwp[k]);
    if (db) atomicAdd(&db[j], dbp[k]);
  }
}

// ---------------------------------------------------------------------------
// Host wrappers
// ---------------------------------------------------------------------------
static int norm_grid(int rows) { return std::min(rows, 2048); }

std::vector<at::Tensor> norm_fwd(at::Tensor x, at::Tensor w,
                                 c10::optional<at::Tensor> b, double eps,
                                 bool rms) {
  TORCH_CHECK(x.is_cuda() && x.is_contiguous());
  const int cols = (int)x.size(-1);
  const int rows = (int)(x.numel() / cols);
  TORCH_CHECK(cols <= 64 * LN_BLOCK, "row too long for norm kernels");
  auto y = at::empty_like(x);
  auto opts = x.options().dtype(at::kFloat);
  auto mean = rms ? at::Tensor() : at::empty({rows}, opts);
  auto rstd = at::empty({rows}, opts);
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(norm_grid(rows)), block(LN_BLOCK);

  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "norm_fwd", [&] {
        using T = scalar_t;
        auto k = rms ? norm_fwd_kernel<T, true> : norm_fwd_kernel<T, false>;
        hipLaunchKernelGGL(k, grid, block, 0, stream.stream(),
                           reinterpret_cast<const T*>(x.data_ptr()),
                           reinterpret_cast<const T*>(w.data_ptr()),
                           b.has_value()
                               ? reinterpret_cast<const T*>(b->data_ptr())
                               : nullptr,
                           reinterpret_cast<T*>(y.data_ptr()),
                           rms ? nullptr : mean.data_ptr<float>(),
                           rstd.data_ptr<float>(), rows, cols, (float)eps);
      });
  return {y, mean, rstd};
}

std::vector<at::Tensor> norm_bwd(at::Tensor dy, at::Tensor x, at::Tensor w,
                                 at::Tensor mean, at::Tensor rstd, bool rms,
                                 bool needs_db) {
  TORCH_CHECK(dy.is_cuda() && dy.is_contiguous() && x.is_contiguous());
  const int cols = (int)x.size(-1);
  const int rows = (int)(x.numel() / cols);
  auto dx = at::empty_like(x);
  auto opts = x.options().dtype(at::kFloat);
  auto dw = at::zeros({cols}, opts);
  auto db = needs_db ? at::zeros({cols}, opts) : at::Tensor();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(norm_grid(rows)), block(LN_BLOCK);

  const int cpt_needed = (cols + LN_BLOCK - 1) / LN_BLOCK;
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "norm_bwd", [&] {
        using T = scalar_t;
        void (*k)(const T*, const T*, const T*, const float*, const float*,
                  T*, float*, float*, int, int);
        if (cpt_needed <= 8)
          k = rms ? norm_bwd_kernel<T, true, 8> : norm_bwd_kernel<T, false, 8>;
        else if (cpt_needed <= 16)
          k = rms ? norm_bwd_kernel<T, true, 16> : norm_bwd_kernel<T, false, 16>;
        else if (cpt_needed <= 32)
          k = rms ? norm_bwd_kernel<T, true, 32> : norm_bwd_kernel<T, false, 32>;
        else
          k = rms ? norm_bwd_kernel<T, true, 64> : norm_bwd_kernel<T, false, 64>;
        hipLaunchKernelGGL(k, grid, block, 0, stream.stream(),
                           reinterpret_cast<const T*>(dy.data_ptr()),
                           reinterpret_cast<const T*>(x.data_ptr()),
                           reinterpret_cast<const T*>(w.data_ptr()),
                           rms ? nullptr : mean.data_ptr<float>(),
                           rstd.data_ptr<float>(),
                           reinterpret_cast<T*>(dx.data_ptr()),
                           dw.data_ptr<float>(),
                           needs_db ? db.data_ptr<float>() : nullptr, rows,
                           cols);
      });
  return {dx, dw, db};
}

}  // namespace samd
