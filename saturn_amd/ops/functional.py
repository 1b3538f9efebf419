"""Autograd wrappers over the CDNA4 kernels, with CPU reference fallbacks.

GPU tensors route to ``saturn_amd._C`` (and raise loudly if the extension is
missing — ops/__init__.require_ext); CPU tensors use the equivalent pure
PyTorch math so the orchestration suite runs in the no-GPU container.  The
numerics tests compare the two paths (tests/test_ops_gpu.py).
"""

from __future__ import annotations

from typing import Tuple

import torch
import torch.nn.functional as F

from saturn_amd.ops import require_ext


# ---------------------------------------------------------------------------
# LayerNorm / RMSNorm (K4)
# ---------------------------------------------------------------------------
class _NormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, w, b, eps, rms):
        ext = require_ext()
        x = x.contiguous()
        y, mean, rstd = ext.norm_fwd(x, w, b, eps, rms)
        if rms:
            mean = rstd  # placeholder; RMS backward ignores it
        ctx.save_for_backward(x, w, mean, rstd)
        ctx.rms = rms
        ctx.has_b = b is not None
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        x, w, mean, rstd = ctx.saved_tensors
        dx, dw, db = ext.norm_bwd(
            dy.contiguous(), x, w, mean, rstd, ctx.rms, ctx.has_b
        )
        return (
            dx,
            dw.to(w.dtype),
            db.to(w.dtype) if ctx.has_b else None,
            None,
            None,
        )


def fused_layer_norm(x, weight, bias=None, eps: float = 1e-5):
    if x.is_cuda:
        return _NormFn.apply(x, weight, bias, eps, False)
    return F.layer_norm(
        x.float(), (x.shape[-1],),
        weight.float(), bias.float() if bias is not None else None, eps
    ).to(x.dtype)


def fused_rms_norm(x, weight, eps: float = 1e-6):
    if x.is_cuda:
        return _NormFn.apply(x, weight, None, eps, True)
    xf = x.float()
    y = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (y * weight.float()).to(x.dtype)


class FusedLayerNorm(torch.nn.Module):
    def __init__(self, dim: int, eps: float = 1e-5, bias: bool = True):
        super().__init__()
        self.eps = eps
        self.weight = torch.nn.Parameter(torch.ones(dim))
        self.bias = torch.nn.Parameter(torch.zeros(dim)) if bias else None

    def forward(self, x):
        return fused_layer_norm(x, self.weight, self.bias, self.eps)


class FusedRMSNorm(torch.nn.Module):
    def __init__(self, dim: int, eps: float = 1e-6):
        super().__init__()
        self.eps = eps
        self.weight = torch.nn.Parameter(torch.ones(dim))

    def forward(self, x):
        return fused_rms_norm(x, self.weight, self.eps)


# ---------------------------------------------------------------------------
# Fused shift + cross-entropy (K8)
# ---------------------------------------------------------------------------
class _CeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets_flat, Tr, ignore_index):
        ext = require_ext()
        loss, lse = ext.ce_fwd(logits, targets_flat, Tr, ignore_index)
        valid = (targets_flat != ignore_index).sum().clamp(min=1)
        ctx.save_for_backward(logits, targets_flat, lse, valid)
        ctx.Tr = Tr
        ctx.ignore_index = ignore_index
        return loss.sum() / valid.float()

    @staticmethod
    def backward(ctx, grad_out):
        ext = require_ext()
        logits, targets_flat, lse, valid = ctx.saved_tensors
        dloss = torch.full_like(lse, 0.0)
        dloss.fill_(1.0)
        dloss = dloss * (grad_out.float() / valid.float())
        dlogits = ext.ce_bwd(
            logits, targets_flat, lse, dloss, ctx.Tr, ctx.ignore_index
        )
        return dlogits, None, None, None


def fused_cross_entropy(
    logits: torch.Tensor,
    targets: torch.Tensor,
    shift: bool = True,
    ignore_index: int = -100,
) -> torch.Tensor:
    """Causal-LM loss: mean CE of ``logits[:, :-1]`` vs ``targets[:, 1:]``
    (``shift=True``) without materializing shifted copies of the logits.

    logits: [B, T, V]; targets: [B, T] int64.
    """
    B, T, V = logits.shape
    if logits.is_cuda:
        Tr = T - 1 if shift else T
        tg = targets[:, 1:] if shift else targets
        return _CeFn.apply(
            logits.contiguous(), tg.contiguous().view(-1), Tr, ignore_index
        )
    # CPU reference
    if shift:
        lg = logits[:, :-1].float().reshape(-1, V)
        tg = targets[:, 1:].reshape(-1)
    else:
        lg = logits.float().reshape(-1, V)
        tg = targets.reshape(-1)
    return F.cross_entropy(lg, tg, ignore_index=ignore_index)


# ---------------------------------------------------------------------------
# Rotary embedding (K3)
# ---------------------------------------------------------------------------
def rope_tables(
    T: int,
    rotary_dim: int,
    base: float = 10000.0,
    device=None,
    interleaved: bool = True,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """cos/sin tables [T, rotary_dim/2] fp32."""
    half = rotary_dim // 2
    inv = 1.0 / (base ** (torch.arange(0, half, dtype=torch.float32) * 2 / rotary_dim))
    t = torch.arange(T, dtype=torch.float32)
    freqs = torch.outer(t, inv)
    return freqs.cos().to(device), freqs.sin().to(device)


class _RopeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, cos_t, sin_t, t_len, t_off, half_style):
        ext = require_ext()
        x = x.contiguous()
        y = torch.empty_like(x)
        ext.rope_apply(y, x, cos_t, sin_t, t_len, t_off, half_style, False)
        ctx.save_for_backward(cos_t, sin_t)
        ctx.meta = (t_len, t_off, half_style)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        cos_t, sin_t = ctx.saved_tensors
        t_len, t_off, half_style = ctx.meta
        dy = dy.contiguous()
        dx = torch.empty_like(dy)
        ext.rope_apply(dx, dy, cos_t, sin_t, t_len, t_off, half_style, True)
        return dx, None, None, None, None, None


def _rope_cpu(x, cos_bt, sin_bt, half_style):
    B, T, H, D = x.shape
    half = cos_bt.shape[-1]
    c = cos_bt.view(B, T, 1, half).float()
    s = sin_bt.view(B, T, 1, half).float()
    xf = x.float()
    out = xf.clone()
    if half_style:
        x0 = xf[..., :half]
        x1 = xf[..., half : 2 * half]
        out[..., :half] = x0 * c - x1 * s
        out[..., half : 2 * half] = x1 * c + x0 * s
    else:
        x0 = xf[..., 0 : 2 * half : 2]
        x1 = xf[..., 1 : 2 * half : 2]
        out[..., 0 : 2 * half : 2] = x0 * c - x1 * s
        out[..., 1 : 2 * half : 2] = x1 * c + x0 * s
    return out.to(x.dtype)


def apply_rope(
    x: torch.Tensor,
    cos: torch.Tensor,
    sin: torch.Tensor,
    half_style: bool = False,
) -> torch.Tensor:
    """x: [B, T, H, D]; cos/sin: fp32 [T_total, half] (broadcast over batch
    and heads — the kernel indexes rows itself, no per-call expand).

    GPT-J uses interleaved pairs (half_style=False, GPTJ.py:56-61);
    Llama/NeoX uses the half-split layout (half_style=True).  Under
    sequence parallelism this rank's shard covers global positions
    [rank*T, rank*T + T) — the kernel reads table rows at that offset.
    """
    B, T, H, D = x.shape
    half = cos.shape[-1]
    from saturn_amd.parallel.sequence import _STATE as _sp

    off = _sp["rank"] * T if _sp["world"] > 1 else 0
    if cos.shape[0] < off + T:
        raise ValueError(
            "rope table too short for this sequence shard — pass the FULL "
            "table to apply_rope (it slices positions itself)"
        )
    if cos.dtype != torch.float32:
        # tables must stay fp32 (model.to(bf16) would quantize angles)
        cos = cos.float()
        sin = sin.float()
    if x.is_cuda:
        return _RopeFn.apply(x, cos.contiguous(), sin.contiguous(), T, off,
                             half_style)
    cos_bt = cos[off : off + T].unsqueeze(0).expand(B, T, half)
    sin_bt = sin[off : off + T].unsqueeze(0).expand(B, T, half)
    return _rope_cpu(x, cos_bt.contiguous(), sin_bt.contiguous(), half_style)


# ---------------------------------------------------------------------------
# Causal attention (K2) — flash kernel lands in attention.hip; the math
# fallback below (explicit GEMMs + softmax, fp32 accum like the reference's
# GPTJ.py:164-191) is used on CPU and as the numerics reference.
# ---------------------------------------------------------------------------
def attention_math(q, k, v, causal: bool = True) -> torch.Tensor:
    """q,k,v: [B, H, T, D].  fp32 score math, returns q.dtype."""
    scale = 1.0 / (q.shape[-1] ** 0.5)
    s = torch.matmul(q.float(), k.float().transpose(-1, -2)) * scale
    if causal:
        T = q.shape[-2]
        mask = torch.ones(T, T, dtype=torch.bool, device=q.device).tril()
        s = s.masked_fill(~mask, float("-inf"))
    p = torch.softmax(s, dim=-1)
    return torch.matmul(p, v.float()).to(q.dtype)


def _attention_core(q, k, v, causal: bool) -> torch.Tensor:
    if q.is_cuda:
        from saturn_amd.ops import flash  # local import: optional kernel

        return flash.flash_attention(q, k, v, causal=causal)
    if k.shape[1] != q.shape[1]:
        rep = q.shape[1] // k.shape[1]
        k = k.repeat_interleave(rep, dim=1)
        v = v.repeat_interleave(rep, dim=1)
    return attention_math(q, k, v, causal=causal)


def _attention(q, k, v, causal: bool) -> torch.Tensor:
    from saturn_amd.parallel.sequence import sp_attention, sp_world

    if sp_world() > 1:
        # Ulysses: all-to-all seq-shard <-> head-shard around the full-
        # sequence kernel (exact causal masking on the gathered sequence)
        return sp_attention(q, k, v, lambda a, b, c: _attention_core(a, b, c, causal))
    return _attention_core(q, k, v, causal)


def causal_attention(q, k, v) -> torch.Tensor:
    """Dispatch: fused CDNA4 flash kernel on GPU (when built), math path on
    CPU.  Handles GQA (k/v with fewer heads) by expansion."""
    return _attention(q, k, v, causal=True)


def full_attention(q, k, v) -> torch.Tensor:
    """Bidirectional (encoder) attention — BERT/ViT."""
    return _attention(q, k, v, causal=False)


# ---------------------------------------------------------------------------
# Fused 3-way residual add (K7)
# ---------------------------------------------------------------------------
class _Add3Fn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, a, b, c):
        ext = require_ext()
        return ext.add3(a.contiguous(), b.contiguous(), c.contiguous())

    @staticmethod
    def backward(ctx, g):
        return g, g, g


def fused_add3(a, b, c):
    """out = a + b + c in one HBM pass (GPT-J parallel block residual)."""
    if a.is_cuda:
        return _Add3Fn.apply(a, b, c)
    return a + b + c


class _GeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        ext = require_ext()
        x = x.contiguous()
        ctx.save_for_backward(x)
        return ext.gelu_fwd(x)

    @staticmethod
    def backward(ctx, dy):
        ext = require_ext()
        (x,) = ctx.saved_tensors
        return ext.gelu_bwd(dy.contiguous(), x)


def fused_gelu(x: torch.Tensor) -> torch.Tensor:
    """tanh-approx GELU (GPT-J's activation, GPTJ.py:25-41) as one
    vectorized CDNA4 kernel each way; torch fallback on CPU / fp32."""
    if x.is_cuda and x.dtype in (torch.bfloat16, torch.float16):
        return _GeluFn.apply(x)
    return torch.nn.functional.gelu(x, approximate="tanh")


class _SwigluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        ext = require_ext()
        gate = gate.contiguous()
        up = up.contiguous()
        ctx.save_for_backward(gate, up)
        return ext.swiglu_fwd(gate, up)

    @staticmethod
    def backward(ctx, dout):
        ext = require_ext()
        gate, up = ctx.saved_tensors
        dgate, dup = ext.swiglu_bwd(dout.contiguous(), gate, up)
        return dgate, dup


def fused_swiglu(gate, up):
    """silu(gate) * up in one pass each way (Llama FFN)."""
    if gate.is_cuda:
        return _SwigluFn.apply(gate, up)
    return torch.nn.functional.silu(gate) * up


class _EmbedFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, weight, idx):
        ext = require_ext()
        ctx.save_for_backward(idx)
        ctx.vocab = weight.shape[0]
        return ext.embed_fwd(weight, idx)

    @staticmethod
    def backward(ctx, dout):
        ext = require_ext()
        (idx,) = ctx.saved_tensors
        return ext.embed_bwd(dout, idx, ctx.vocab), None


def fused_embedding(weight, idx):
    """Token embedding gather; backward is a fused fp32-atomic scatter-add
    into the vocab table (reference GPTJ.py:346,377)."""
    if weight.is_cuda:
        return _EmbedFn.apply(weight, idx)
    return F.embedding(idx, weight)


class FusedEmbedding(torch.nn.Embedding):
    """Drop-in ``nn.Embedding`` that routes to the K5 HIP kernels on GPU.

    Subclassing keeps state-dict keys, init-weight isinstance checks and TP
    sharding untouched; exotic Embedding options (padding_idx, max_norm,
    sparse) fall back to the stock op.
    """

    def forward(self, idx):  # noqa: D102
        if (
            self.weight.is_cuda
            and self.padding_idx is None
            and self.max_norm is None
            and not self.sparse
            and self.weight.dtype
            in (torch.bfloat16, torch.float16, torch.float32)
        ):
            return _EmbedFn.apply(self.weight, idx)
        return super().forward(idx)


class _DropoutFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, p, seed):
        ext = require_ext()
        ctx.p = p
        ctx.seed = seed
        return ext.dropout_fwd(x, p, seed)

    @staticmethod
    def backward(ctx, dout):
        ext = require_ext()
        return ext.dropout_bwd(dout.contiguous(), ctx.p, ctx.seed), None, None


def fused_dropout(x, p: float, training: bool = True):
    """Counter-based dropout: the backward regenerates the keep
    mask from a 64-bit seed instead of storing a mask tensor, so both passes
    stay single-read/single-write HBM-bound (reference GPTJ.py:95-96,347)."""
    if p == 0.0 or not training:
        return x
    if x.is_cuda and x.dtype in (torch.bfloat16, torch.float16):
        seed = int(torch.randint(0, 2**62, (1,)).item())
        return _DropoutFn.apply(x.contiguous(), p, seed)
    return F.dropout(x, p, training)


class FusedDropout(torch.nn.Module):
    def __init__(self, p: float = 0.0):
        super().__init__()
        self.p = p

    def forward(self, x):  # noqa: D102
        return fused_dropout(x, self.p, self.training)

    def extra_repr(self):  # noqa: D102
        return f"p={self.p}"
