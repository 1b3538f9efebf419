"""Flash attention dispatch (K2).

``saturn_amd._C.attn_fwd/attn_bwd`` are the hand-written CDNA4 MFMA kernels
(attention.hip).  Until they are built into the extension, the GPU path
falls back to the explicit-GEMM math composition (rocBLAS GEMMs + softmax —
no Triton, no aotriton) and WARNS once: measurements taken on the fallback
are not this framework's attention numbers.
"""

from __future__ import annotations

import logging

import torch

from saturn_amd.ops import require_ext

log = logging.getLogger(__name__)
_warned = False


def _pad_t64(t: torch.Tensor) -> torch.Tensor:
    """Zero-pad the T dim (dim -2) up to a multiple of 64."""
    pad = (-t.shape[-2]) % 64
    if pad == 0:
        return t
    return torch.nn.functional.pad(t, (0, 0, 0, pad))


class _FlashFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal):
        ext = require_ext()
        T = q.shape[-2]
        if T % 64:
            # ragged T (ViT's 197, BERT's raw lengths): zero-pad to the
            # kernels' 64-row tiles; padded KEYS are masked in-kernel via
            # kv_len and padded q rows are sliced off below (their zero
            # dO rows contribute exactly 0 to dK/dV in backward)
            q, k, v = _pad_t64(q), _pad_t64(k), _pad_t64(v)
        o, lse = ext.attn_fwd(q, k, v, causal, T)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        ctx.t_real = T
        return o[:, :, :T] if T % 64 else o

    @staticmethod
    def backward(ctx, do):
        q, k, v, o, lse = ctx.saved_tensors
        ext = require_ext()
        T = ctx.t_real
        if hasattr(ext, "attn_bwd"):
            do = _pad_t64(_dense_rows(do)) if T % 64 else _dense_rows(do)
            dq, dk, dv = ext.attn_bwd(do, q, k, v, o, lse, ctx.causal, T)
            if T % 64:
                dq, dk, dv = dq[:, :, :T], dk[:, :, :T], dv[:, :, :T]
            return dq, dk, dv, None
        # Analytic FA2 backward composed from rocBLAS GEMMs (recompute P
        # from the saved LSE; O(T^2) transient, fp32 math).  The fused HIP
        # backward kernel replaces this path when built.
        scale = 1.0 / (q.shape[-1] ** 0.5)
        qf, kf, vf = q.float(), k.float(), v.float()
        dof, of = do.float(), o.float()
        s = torch.matmul(qf, kf.transpose(-1, -2)) * scale
        if ctx.causal:
            T = q.shape[-2]
            mask = torch.ones(T, T, dtype=torch.bool, device=q.device).tril()
            s = s.masked_fill(~mask, float("-inf"))
        p = torch.exp(s - lse.unsqueeze(-1))
        dv = torch.matmul(p.transpose(-1, -2), dof)
        dp = torch.matmul(dof, vf.transpose(-1, -2))
        delta = (dof * of).sum(-1, keepdim=True)
        ds = p * (dp - delta) * scale
        dq = torch.matmul(ds, kf)
        dk = torch.matmul(ds.transpose(-1, -2), qf)
        return dq.to(q.dtype), dk.to(k.dtype), dv.to(v.dtype), None


def _kernel_supported(q) -> bool:
    # ragged T is padded to 64 rows by the wrapper; D stays restricted to
    # the MFMA tile shapes
    return q.dtype == torch.bfloat16 and q.shape[-1] in (64, 128, 256)


def _dense_rows(t):
    """The kernels address arbitrary (B, H, T) strides but need a dense,
    16-B-aligned innermost dim — transposed views of the model's [B,T,H,D]
    projections qualify, so no copies in the common path."""
    if t.stride(-1) == 1 and t.stride(2) % 8 == 0:
        return t
    return t.contiguous()


def flash_attention(q, k, v, causal: bool = True) -> torch.Tensor:
    """q,k,v: [B, H, T, D] bf16/fp16 contiguous (GQA: H_kv may divide H)."""
    ext = require_ext()
    if hasattr(ext, "attn_fwd") and _kernel_supported(q):
        # GQA is native in the kernels: kv may keep fewer heads
        return _FlashFn.apply(
            _dense_rows(q), _dense_rows(k), _dense_rows(v), causal
        )
    global _warned
    if not _warned:
        log.warning(
            "attn_fwd not in saturn_amd._C — using explicit-GEMM math "
            "attention (slower; rebuild with attention.hip)"
        )
        _warned = True
    from saturn_amd.ops.functional import attention_math

    if k.shape[1] != q.shape[1]:
        rep = q.shape[1] // k.shape[1]
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    return attention_math(q, k, v, causal=causal)
