"""Fused multi-tensor optimizers.

On GPU these call the hand-written CDNA4 multi-tensor-apply kernels in
``saturn_amd._C`` (one launch updates every parameter chunk; HBM-bound, so
the kernel is a vectorized grid-stride sweep).  On CPU they fall back to
fp32-accumulating torch math so the orchestration test suite runs in the
no-GPU container with the same numerics contract as the kernel:

- momentum buffers and Adam moments are ALWAYS fp32 regardless of the
  parameter dtype (the HIP kernel requires it);
- ``master_weights=True`` keeps an fp32 master copy per low-precision
  parameter: the fused kernel updates the master in fp32 and writes the
  rounded bf16/fp16 copy in the same launch, avoiding long-horizon update
  cancellation when ``lr * grad`` drops below bf16 resolution.  The
  reference trains fp32 models (simple-verification.py:59) and never had
  this problem.
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
        master_weights: bool = False,
    ) -> None:
        defaults = dict(
            lr=lr,
            momentum=momentum,
            weight_decay=weight_decay,
            master_weights=master_weights,
        )
        super().__init__(params, defaults)

    def _master_of(self, p: torch.nn.Parameter) -> Optional[torch.Tensor]:
        st = self.state[p]
        if "master" not in st:
            st["master"] = p.data.detach().to(torch.float32).contiguous()
        return st["master"]

    @torch.no_grad()
    def step(self, closure=None) -> Optional[float]:
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            lr = group["lr"]
            mom = group["momentum"]
            wd = group["weight_decay"]
            use_master = group.get("master_weights", False)
            ps = [p for p in group["params"] if p.grad is not None]
            for (device, dtype), chunk in _grouped(ps).items():
                grads = [p.grad for p in chunk]
                if mom != 0.0:
                    bufs = []
                    for p in chunk:
                        st = self.state[p]
                        if "momentum_buffer" not in st:
                            # fp32 regardless of param dtype: the HIP kernel
                            # requires fp32 momentum (fused_optim.hip) and
                            # low-precision momentum accumulation diverges
                            st["momentum_buffer"] = torch.zeros_like(
                                p, dtype=torch.float32
                            )
                        bufs.append(st["momentum_buffer"])
                else:
                    bufs = None
                masters = None
                if use_master and dtype != torch.float32:
                    masters = [self._master_of(p) for p in chunk]
                if device.type == "cuda":
                    ext = require_ext()
                    ext.fused_sgd(
                        [p.data for p in chunk],
                        grads,
                        bufs if bufs is not None else [],
                        masters if masters is not None else [],
                        lr,
                        mom,
                        wd,
                    )
                else:
                    # fp32-accumulating fallback, same math as the kernel
                    for i, p in enumerate(chunk):
                        g32 = grads[i].float()
                        pv = (
                            masters[i]
                            if masters is not None
                            else p.data.float()
                        )
                        if wd != 0.0:
                            g32 = g32.add(pv, alpha=wd)
                        if bufs is not None:
                            bufs[i].mul_(mom).add_(g32)
                            g32 = bufs[i]
                        nv = pv.add_(g32, alpha=-lr) if masters is not None \
                            else pv.add(g32, alpha=-lr)
                        p.data.copy_(nv.to(p.dtype))
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
        master_weights: bool = False,
    ) -> None:
        defaults = dict(
            lr=lr,
            betas=betas,
            eps=eps,
            weight_decay=weight_decay,
            master_weights=master_weights,
        )
        super().__init__(params, defaults)

    @torch.no_grad()
    def step(self, closure=None) -> Optional[float]:
        loss = closure() if closure is not None else None
        for group in self.param_groups:
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps = group["eps"]
            wd = group["weight_decay"]
            use_master = group.get("master_weights", False)
            ps = [p for p in group["params"] if p.grad is not None]
            for (device, dtype), chunk in _grouped(ps).items():
                m, v, steps = [], [], []
                masters = [] if use_master and dtype != torch.float32 else None
                for p in chunk:
                    st = self.state[p]
                    if "step" not in st:
                        st["step"] = 0
                        # fp32 moments regardless of param dtype (bf16-safe)
                        st["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
                        st["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
                    if masters is not None and "master" not in st:
                        st["master"] = p.data.detach().to(torch.float32).contiguous()
                    st["step"] += 1
                    m.append(st["exp_avg"])
                    v.append(st["exp_avg_sq"])
                    if masters is not None:
                        masters.append(st["master"])
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
                        masters if masters is not None else [],
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
                    for i, (p, mi, vi) in enumerate(zip(chunk, m, v)):
                        denom = (vi / bc2).sqrt_().add_(eps)
                        upd = (mi / bc1).div_(denom)
                        pv = masters[i] if masters is not None else p.data.float()
                        if wd != 0.0:
                            pv.mul_(1 - lr * wd)
                        pv.add_(upd, alpha=-lr)
                        p.data.copy_(pv.to(p.dtype))
        return loss
