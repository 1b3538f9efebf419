// Fused causal-LM cross-entropy over large vocab.
//
// Replaces the reference's shift + nn.CrossEntropyLoss pair
// (GPTJ.py:491-499): no materialized log-probs, single online-softmax pass
// over the 50k vocab per row in the forward (per-thread running max/sum,
// block merge), one recompute pass in the backward that writes
// grad-logits directly in the IO dtype.
//
// Rows address a [B, T, V] logits tensor through (batch_stride, Tr) so the
// shifted view logits[:, :-1, :] needs no contiguous copy.

#include <torch/extension.h>
#include <ATen/hip/HIPContext.h>

#include "common.h"

namespace samd {

constexpr int CE_BLOCK = 256;

template <typename T>
__global__ void ce_fwd_kernel(const T* __restrict__ logits,
                              const long* __restrict__ targets,
                              float* __restrict__ loss,
                              float* __restrict__ lse_out, int n_rows, int Tr,
                              long batch_stride, int V, long ignore_index) {
  __shared__ float lds[CE_BLOCK / WAVE];
  for (int row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const long tgt = targets[row];
    const T* xr =
        logits + (long)(row / Tr) * batch_stride + (long)(row % Tr) * V;
    if (tgt == ignore_index) {
      if (threadIdx.x == 0) {
        loss[row] = 0.f;
        lse_out[row] = 0.f;
      }
      __syncthreads();
      continue;
    }
    // per-thread online max/sum over a strided slice
    float m = -INFINITY, s = 0.f;
    for (int j = threadIdx.x; j < V; j += CE_BLOCK) {
      float v = (float)xr[j];
      if (v > m) {
        s = s * __expf(m - v) + 1.f;
        m = v;
      } else {
        s += __expf(v - m);
      }
    }
    // block merge: global max, then rescaled sums
    float gm = block_max<CE_BLOCK>(m, lds);
    float part = (m == -INFINITY) ? 0.f : s * __expf(m - gm);
    float gs = block_sum<CE_BLOCK>(part, lds);
    if (threadIdx.x == 0) {
      float lse = gm + __logf(gs);
      lse_out[row] = lse;
      loss[row] = lse - (float)xr[tgt];
    }
    __syncthreads();
  }
}

template <typename T>
__global__ void ce_bwd_kernel(const T* __restrict__ logits,
                              const long* __restrict__ targets,
                              const float* __restrict__ lse,
                              const float* __restrict__ dloss,
                              T* __restrict__ dlogits, int n_rows, int Tr,
                              long batch_stride, int V, long ignore_index) {
  for (int row = blockIdx.x; row < n_rows; row += gridDim.x) {
    const long tgt = targets[row];
    const long off =
        (long)(row / Tr) * batch_stride + (long)(row % Tr) * V;
    const T* xr = logits + off;
    T* dxr = dlogits + off;
    if (tgt == ignore_index) {
      for (int j = threadIdx.x; j < V; j += CE_BLOCK) dxr[j] = (T)0.f;
      continue;
    }
    const float l = lse[row];
    const float scale = dloss[row];
    for (int j = threadIdx.x; j < V; j += CE_BLOCK) {
      float p = __expf((float)xr[j] - l);
      float g = scale * (p - (j == (int)tgt ? 1.f : 0.f));
      dxr[j] = (T)g;
    }
  }
}

std::vector<at::Tensor> ce_fwd(at::Tensor logits, at::Tensor targets, int64_t Tr,
                               int64_t ignore_index) {
  // logits: [B, T, V] contiguous; rows = B * Tr taken from the first Tr
  // positions of each sequence; targets: [B*Tr] contiguous int64.
  TORCH_CHECK(logits.is_cuda() && logits.dim() == 3 && logits.is_contiguous());
  TORCH_CHECK(targets.scalar_type() == at::kLong && targets.is_contiguous());
  const int B = (int)logits.size(0);
  const int V = (int)logits.size(2);
  const int n_rows = (int)(B * Tr);
  TORCH_CHECK(targets.numel() == n_rows);
  auto opts = logits.options().dtype(at::kFloat);
  auto loss = at::empty({n_rows}, opts);
  auto lse = at::empty({n_rows}, opts);
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(std::min(n_rows, 2048)), block(CE_BLOCK);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, logits.scalar_type(), "ce_fwd", [&] {
        hipLaunchKernelGGL(ce_fwd_kernel<scalar_t>, grid, block, 0,
                           stream.stream(),
                           reinterpret_cast<const scalar_t*>(logits.data_ptr()),
                           targets.data_ptr<long>(), loss.data_ptr<float>(),
                           lse.data_ptr<float>(), n_rows, (int)Tr,
                           (long)logits.stride(0), V, ignore_index);
      });
  return {loss, lse};
}

at::Tensor ce_bwd(at::Tensor logits, at::Tensor targets, at::Tensor lse,
                  at::Tensor dloss, int64_t Tr, int64_t ignore_index) {
  const int B = (int)logits.size(0);
  const int V = (int)logits.size(2);
  const int n_rows = (int)(B * Tr);
  const int T = (int)logits.size(1);
  // Gradient for the full [B, T, V] tensor; the kernel writes rows < Tr,
  // only the (T - Tr) tail rows need zeroing (avoids a full 2x write of
  // the 400+ MB grad tensor that zeros_like would cost).
  auto dlogits = at::empty_like(logits);
  if (T > Tr) dlogits.narrow(1, Tr, T - Tr).zero_();
  auto stream = at::hip::getCurrentHIPStream();
  dim3 grid(std::min(n_rows, 2048)), block(CE_BLOCK);
  AT_DISPATCH_FLOATING_TYPES_AND2(
      at::kBFloat16, at::kHalf, logits.scalar_type(), "ce_bwd", [&] {
        hipLaunchKernelGGL(ce_bwd_kernel<scalar_t>, grid, block, 0,
                           stream.stream(),
                           reinterpret_cast<const scalar_t*>(logits.data_ptr()),
                           targets.data_ptr<long>(), lse.data_ptr<float>(),
                           dloss.data_ptr<float>(),
                           reinterpret_cast<scalar_t*>(dlogits.data_ptr()),
                           n_rows, (int)Tr, (long)logits.stride(0), V,
                           ignore_index);
      });
  return dlogits;
}

}  // namespace samd
