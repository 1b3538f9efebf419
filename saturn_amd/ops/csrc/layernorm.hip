// LayerNorm + RMSNorm forward/backward for CDNA4.
//
// One 256-thread block (4 waves) per chunk of rows; each thread owns a
// fixed column slice so the backward's dweight/dbias partials accumulate in
// registers across the block's rows and hit global memory once per block
// (guide Guideline 12: per-block partial reduction before atomics).
// Stats are fp32; IO is bf16/fp16/fp32.
//
// Round-2 rework: the round-1 kernels walked columns with a 2-byte scalar
// load per thread (G13: ~2-2.5x cost) and re-read the row on the second
// pass; measured 60/165 us per 8192x4096 call vs ~17/25 us of pure HBM
// traffic.  16-bit dtypes with cols % 8 == 0 now load 16-B groups (8
// elements) per thread and cache the row in registers between passes.
//
// Replaces the implicit torch kernels behind nn.LayerNorm at reference
// GPTJ.py:396,350 with hand-written wave64 kernels.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace samd {

constexpr int LN_BLOCK = 256;

typedef __attribute__((ext_vector_type(8))) short short8v_ln;

template <typename T>
__device__ __forceinline__ float elemF(const short8v_ln& v, int e) {
  T t;
  t.x = (unsigned short)v[e];
  return toF<T>(t);
}

template <typename T>
__device__ __forceinline__ void setElem(short8v_ln& v, int e, float f) {
  v[e] = (short)fromF<T>(f).x;
}

// ---------------------------------------------------------------------------
// Forward: y = (x - mean) * rstd * w + b     (RMS: y = x * rrms * w)
// ---------------------------------------------------------------------------

// Vector path: 16-bit T, cols % 8 == 0, cols <= 8 * LN_BLOCK * GMAX.
// GPT (GMAX=2) covers cols <= 4096; GMAX=8 covers 16384 (llama-70B ffn).
template <typename T, bool RMS, int GMAX>
__global__ void norm_fwd_vec_kernel(const T* __restrict__ x,
                                    const T* __restrict__ w,
                                    const T* __restrict__ b, T* __restrict__ y,
                                    float* __restrict__ mean_out,
                                    float* __restrict__ rstd_out, int rows,
                                    int cols, float eps) {
  __shared__ float lds[LN_BLOCK / WAVE];
  // GMAX is a compile-time bound and the loops fully unroll: a runtime
  // trip count would turn cache[] indexing dynamic and demote the arrays
  // to scratch (measured 5x regression).
  float cache[8 * GMAX];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* xr = x + (long)row * cols;
    T* yr = y + (long)row * cols;
    float s = 0.f, sq = 0.f;
#pragma unroll
    for (int g = 0; g < GMAX; ++g) {
      const int j = (g * LN_BLOCK + threadIdx.x) * 8;
      if (j < cols) {
        const short8v_ln xv = *reinterpret_cast<const short8v_ln*>(xr + j);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float v = elemF<T>(xv, e);
          cache[g * 8 + e] = v;
          s += v;
          sq += v * v;
        }
      }
    }
    float mean = 0.f;
    if (!RMS) mean = block_sum<LN_BLOCK>(s, lds) / cols;
    const float var = block_sum<LN_BLOCK>(sq, lds) / cols - mean * mean;
    const float rstd = rsqrtf(var + eps);
    if (threadIdx.x == 0) {
      if (!RMS && mean_out) mean_out[row] = mean;
      if (rstd_out) rstd_out[row] = rstd;
    }
#pragma unroll
    for (int g = 0; g < GMAX; ++g) {
      const int j = (g * LN_BLOCK + threadIdx.x) * 8;
      if (j < cols) {
        const short8v_ln wv = *reinterpret_cast<const short8v_ln*>(w + j);
        short8v_ln bv;
        if (!RMS && b) bv = *reinterpret_cast<const short8v_ln*>(b + j);
        short8v_ln ov;
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float v = (cache[g * 8 + e] - mean) * rstd;
          float o = v * elemF<T>(wv, e);
          if (!RMS && b) o += elemF<T>(bv, e);
          setElem<T>(ov, e, o);
        }
        *reinterpret_cast<short8v_ln*>(yr + j) = ov;
      }
    }
    __syncthreads();
  }
}

// Scalar fallback (any dtype / cols).
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

// Vector path: 16-B loads, row cached in registers between the reduction
// and dx passes, dw/db partials per owned column group.
template <typename T, bool RMS, int GMAX>
__global__ void norm_bwd_vec_kernel(const T* __restrict__ dy,
                                    const T* __restrict__ x,
                                    const T* __restrict__ w,
                                    const float* __restrict__ mean,
                                    const float* __restrict__ rstd,
                                    T* __restrict__ dx, float* __restrict__ dw,
                                    float* __restrict__ db, int rows,
                                    int cols) {
  // dw/db here are PER-BLOCK partial rows [gridDim.x][cols], reduced by
  // norm_bwd_reduce_kernel: every column is hit by every block, so
  // atomicAdd serialized gridDim-deep per address (measured ~460 us of a
  // 485 us call at 8192x4096 — the round-1 kernel paid the same tail).
  __shared__ float lds[LN_BLOCK / WAVE];
  // see norm_fwd_vec_kernel: loops must unroll over the GMAX bound so the
  // register arrays stay in registers
  float dwp[8 * GMAX];
  float dbp[8 * GMAX];
#pragma unroll
  for (int k = 0; k < 8 * GMAX; ++k) { dwp[k] = 0.f; dbp[k] = 0.f; }
  // row cache only at GMAX<=2: at GMAX=4 the four arrays are 128 floats
  // and the kernel spills ~200 B/lane to scratch (compiler-verified) —
  // re-reading the row from HBM is cheaper than scratch traffic
  constexpr bool CACHE = GMAX <= 2;
  float c_dyw[CACHE ? 8 * GMAX : 1];
  float c_xh[CACHE ? 8 * GMAX : 1];
  for (int row = blockIdx.x; row < rows; row += gridDim.x) {
    const T* dyr = dy + (long)row * cols;
    const T* xr = x + (long)row * cols;
    T* dxr = dx + (long)row * cols;
    const float mu = RMS ? 0.f : mean[row];
    const float rs = rstd[row];
    float s1 = 0.f, s2 = 0.f;
#pragma unroll
    for (int g = 0; g < GMAX; ++g) {
      const int j = (g * LN_BLOCK + threadIdx.x) * 8;
      if (j < cols) {
        const short8v_ln dyv = *reinterpret_cast<const short8v_ln*>(dyr + j);
        const short8v_ln xv = *reinterpret_cast<const short8v_ln*>(xr + j);
        const short8v_ln wv = *reinterpret_cast<const short8v_ln*>(w + j);
#pragma unroll
        for (int e = 0; e < 8; ++e) {
          const float dyf = elemF<T>(dyv, e);
          const float xh = (elemF<T>(xv, e) - mu) * rs;
          const float dyw = dyf * elemF<T>(wv, e);
          if constexpr (CACHE) {
            c_dyw[g * 8 + e] = dyw;
            c_xh[g * 8 + e] = xh;
          }
          s1 += dyw;
          s2 += dyw * xh;
          dwp[g * 8 + e] += dyf * xh;
          dbp[g * 8 + e] += dyf;
        }
      }
    }
    const float m1 = RMS ? 0.f : block_sum<LN_BLOCK>(s1, lds) / cols;
    const float m2 = block_sum<LN_BLOCK>(s2, lds) / cols;
#pragma unroll
    for (int g = 0; g < GMAX; ++g) {
      const int j = (g * LN_BLOCK + threadIdx.x) * 8;
      if (j < cols) {
        short8v_ln ov;
        if constexpr (CACHE) {
#pragma unroll
          for (int e = 0; e < 8; ++e)
            setElem<T>(ov, e,
                       rs * (c_dyw[g * 8 + e] - m1 - c_xh[g * 8 + e] * m2));
        } else {
          const short8v_ln dyv = *reinterpret_cast<const short8v_ln*>(dyr + j);
          const short8v_ln xv = *reinterpret_cast<const short8v_ln*>(xr + j);
          const short8v_ln wv = *reinterpret_cast<const short8v_ln*>(w + j);
#pragma unroll
          for (int e = 0; e < 8; ++e) {
            const float dyw = elemF<T>(dyv, e) * elemF<T>(wv, e);
            const float xh = (elemF<T>(xv, e) - mu) * rs;
            setElem<T>(ov, e, rs * (dyw - m1 - xh * m2));
          }
        }
        *reinterpret_cast<short8v_ln*>(dxr + j) = ov;
      }
    }
    __syncthreads();
  }
  // coalesced partial-row stores: thread t owns cols [8t, 8t+8)
  float* dwr = dw + (long)blockIdx.x * cols;
  float* dbr = db ? db + (long)blockIdx.x * cols : nullptr;
#pragma unroll
  for (int g = 0; g < GMAX; ++g) {
    const int j = (g * LN_BLOCK + threadIdx.x) * 8;
    if (j < cols) {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        dwr[j + e] = dwp[g * 8 + e];
        if (db) dbr[j + e] = dbp[g * 8 + e];
      }
    }
  }
}

// The per-block partial rows are reduced with at::sum (host side): a
// hand-rolled one-thread-per-column loop was latency-bound at 16
// workgroups (measured 472 us vs ~15 us for torch's tree reduction).

// Scalar fallback.
template <typename T, bool RMS, int CPT>
__global__ void norm_bwd_kernel(const T* __restrict__ dy, const T* __restrict__ x,
                                const T* __restrict__ w,
                                const float* __restrict__ mean,
                                const float* __restrict__ rstd,
                                T* __restrict__ dx, float* __restrict__ dw,
                                float* __restrict__ db, int rows, int cols) {
  __shared__ float lds[LN_BLOCK / WAVE];
  float dwp[CPT];
  float dbp[CPT];
#pragma unroll
  for (int k = 0; k < CPT; ++k) { dwp[k] = 0.f; dbp[k] = 0.f; }

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
  const bool vec_ok = x.scalar_type() != at::kFloat && cols % 8 == 0 &&
                      cols <= 8 * LN_BLOCK * 8;

  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "norm_fwd", [&] {
        using T = scalar_t;
        if constexpr (sizeof(T) == 2) {
          if (vec_ok) {
            const int gneeded = (cols + 8 * LN_BLOCK - 1) / (8 * LN_BLOCK);
            void (*kv)(const T*, const T*, const T*, T*, float*, float*, int,
                       int, float);
            if (gneeded <= 2)
              kv = rms ? norm_fwd_vec_kernel<T, true, 2>
                       : norm_fwd_vec_kernel<T, false, 2>;
            else
              kv = rms ? norm_fwd_vec_kernel<T, true, 8>
                       : norm_fwd_vec_kernel<T, false, 8>;
            hipLaunchKernelGGL(kv, grid, block, 0, stream.stream(),
                               reinterpret_cast<const T*>(x.data_ptr()),
                               reinterpret_cast<const T*>(w.data_ptr()),
                               b.has_value()
                                   ? reinterpret_cast<const T*>(b->data_ptr())
                                   : nullptr,
                               reinterpret_cast<T*>(y.data_ptr()),
                               rms ? nullptr : mean.data_ptr<float>(),
                               rstd.data_ptr<float>(), rows, cols,
                               (float)eps);
            return;
          }
        }
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
  const bool vec_ok =
      x.scalar_type() != at::kFloat && cols % 8 == 0 && cols <= 8 * LN_BLOCK * 4;
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, x.scalar_type(), "norm_bwd", [&] {
        using T = scalar_t;
        if constexpr (sizeof(T) == 2) {
          if (vec_ok) {
            const int gneeded = (cols + 8 * LN_BLOCK - 1) / (8 * LN_BLOCK);
            void (*kv)(const T*, const T*, const T*, const float*,
                       const float*, T*, float*, float*, int, int);
            // GMAX=4 caps the register cache at 2x32 floats + 2x32 partials
            if (gneeded <= 2)
              kv = rms ? norm_bwd_vec_kernel<T, true, 2>
                       : norm_bwd_vec_kernel<T, false, 2>;
            else
              kv = rms ? norm_bwd_vec_kernel<T, true, 4>
                       : norm_bwd_vec_kernel<T, false, 4>;
            const int nblk = grid.x;
            auto dw_part = at::empty({nblk, cols}, opts);
            auto db_part =
                needs_db ? at::empty({nblk, cols}, opts) : at::Tensor();
            hipLaunchKernelGGL(kv, grid, block, 0, stream.stream(),
                               reinterpret_cast<const T*>(dy.data_ptr()),
                               reinterpret_cast<const T*>(x.data_ptr()),
                               reinterpret_cast<const T*>(w.data_ptr()),
                               rms ? nullptr : mean.data_ptr<float>(),
                               rstd.data_ptr<float>(),
                               reinterpret_cast<T*>(dx.data_ptr()),
                               dw_part.data_ptr<float>(),
                               needs_db ? db_part.data_ptr<float>() : nullptr,
                               rows, cols);
            at::sum_out(dw, dw_part, {0});
            if (needs_db) at::sum_out(db, db_part, {0});
            return;
          }
        }
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
