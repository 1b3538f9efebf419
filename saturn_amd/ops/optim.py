"""Fused multi-tensor optimizers.

On GPU these call the hand-written CDNA4 multi-tensor-apply kernels in
``saturn_amd._C`` (one launch updates every parameter chunk; HBM-bound, so
the kernel is a vectorized grid-stride sweep).  On CPU they fall back to
torch ``_foreach_`` ops so the orchestration test suite runs in the no-GPU
container.  The reference uses stock ``torch.optim.SGD``
(simple-verification.py:59); fused Adam is required by the north star.
"""

from __future__ import annotations


from typing import Iterable, List, Optional

import torch

from saturn_amd.ops import require_ext


def _grouped(params_with_grads: List[torch.nn.Parameter]):
    """Group params by (device, dtype) for multi-tensor apply."""
    groups = {}
    for p in params_with_grads:
        groups.setdefault((p.device, p.dtype), []).append(p)
    return groups


class FusedSGD(torch.optim.Optimizer):
    """SGD with optional momentum/weight decay, fused on GPU."""

    def __init__(
        self,
        params: Iterable[torch.nn.Parameter],
        lr: float,
        momentum: float = 0.0,
        weight_decay: float = 0.0,
    ) -> None:
        defaults = dict(lr=lr, momentum=momentum, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None) -> Optional[float]:
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            lr = group["lr"]
            mom = group["momentum"]
            wd = group["weight_decay"]
            ps = [p for p in group["params"] if p.grad is not None]
            for (device, _dtype), chunk in _grouped(ps).items():
                grads = [p.grad for p in chunk]
                if mom != 0.0:
                    bufs = []
                    for p in chunk:
                        st = self.state[p]
                        if "momentum_buffer" not in st:
                            st["momentum_buffer"] = torch.zeros_like(p)
                        bufs.append(st["momentum_buffer"])
                else:
                    bufs = None
                if device.type == "cuda":
                    ext = require_ext()
                    ext.fused_sgd(
                        [p.data for p in chunk],
                        grads,
                        bufs if bufs is not None else [],
                        lr,
                        mom,
                        wd,
                    )
                else:
                    if wd != 0.0:
                        torch._foreach_add_(
                            grads, [p.data for p in chunk], alpha=wd
                        )
                    if bufs is not None:
                        torch._foreach_mul_(bufs, mom)
                        torch._foreach_add_(bufs, grads)
                        grads = bufs
                    torch._foreach_add_(
                        [p.data for p in chunk], grads, alpha=-lr
                    )
        return loss


class FusedAdam(torch.optim.Optimizer):
    """AdamW-style fused optimizer (decoupled weight decay)."""

    def __init__(
        self,
        params: Iterable[torch.nn.Parameter],
        lr: float = 1e-3,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.0,
    ) -> None:
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None) -> Optional[float]:
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps = group["eps"]
            wd = group["weight_decay"]
            ps = [p for p in group["params"] if p.grad is not None]
            for (device, _dtype), chunk in _grouped(ps).items():
                m, v, steps = [], [], []
                for p in chunk:
                    st = self.state[p]
                    if "step" not in st:
                        st["step"] = 0
                        # fp32 moments regardless of param dtype (bf16-safe)
                        st["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                        st["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                    st["step"] += 1
                    m.append(st["exp_avg"])
                    v.append(st["exp_avg_sq"])
                    steps.append(st["step"])
                step_t = steps[0]  # uniform within a chunk in practice
                bc1 = 1.0 - beta1**step_t
                bc2 = 1.0 - beta2**step_t
                grads = [p.grad for p in chunk]
                if device.type == "cuda":
                    ext = require_ext()
                    ext.fused_adam(
                        [p.data for p in chunk],
                        grads,
                        m,
                        v,
                        lr,
                        beta1,
                        beta2,
                        eps,
                        wd,
                        bc1,
                        bc2,
                    )
                else:
                    gf = [g.float() for g in grads]
                    torch._foreach_mul_(m, beta1)
                    torch._foreach_add_(m, gf, alpha=1 - beta1)
                    torch._foreach_mul_(v, beta2)
                    torch._foreach_addcmul_(v, gf, gf, value=1 - beta2)
                    for p, mi, vi in zip(chunk, m, v):
                        denom = (vi / bc2).sqrt_().add_(eps)
                        upd = (mi / bc1).div_(denom)
                        if wd != 0.0:
                            p.data.mul_(1 - lr * wd)
                        p.data.add_(upd.to(p.dtype), alpha=-lr)
        return loss
