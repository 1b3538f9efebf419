"""hipGraph-captured training step.

``torch.cuda.CUDAGraph`` on ROCm is hipGraph capture/replay.  A whole
fwd+bwd+optimizer step recorded once and replayed per batch removes the
per-kernel launch round-trips — the win is for launch-bound work: small
models, short sequences, and the trial runner's rapid-fire profiling steps
(hundreds of sub-100-us kernels per step).

The framework's training stack is graph-safe by construction:

- ``BucketedDDP`` grads are views into persistent flat buckets
  (``parallel/ddp.py``) — no per-step allocation, ``zero_grad_buffers``
  zeroes in place;
- the fused optimizers (``ops/optim.py``) update params in place with
  pointer tables fixed at capture;
- the fused kernels never sync with the host mid-step.

Constraints (documented, asserted where cheap): the input batch must be
copied into the static buffer (``set_input``) before ``replay``; host-side
RNG (e.g. the dropout seed draw) is executed once at capture, so dropout
masks are frozen under replay — capture with p=0 or accept a fixed mask.
Collectives inside graphs are not exercised this round: use world_size 1.
"""

from __future__ import annotations

from typing import Callable, Optional

import torch


class GraphedStep:
    """Capture ``step_fn`` (fwd+bwd+optimizer, reading only static buffers)
    into a hipGraph; ``replay()`` runs the whole step as one graph launch.

    ``step_fn`` must return the loss tensor; the same storage is reused every
    replay, so read it (``.loss``) before the next ``replay``.
    """

    def __init__(
        self,
        step_fn: Callable[[], torch.Tensor],
        warmup: int = 3,
        stream: Optional[torch.cuda.Stream] = None,
    ):
        assert torch.cuda.is_available(), "GraphedStep needs a GPU"
        self._graph = torch.cuda.CUDAGraph()
        s = stream or torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(max(1, warmup)):
                step_fn()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        with torch.cuda.graph(self._graph):
            self.loss = step_fn()

    def replay(self) -> torch.Tensor:
        self._graph.replay()
        return self.loss


def graphed_train_step(model, loss_fn, optimizer, example_input, ddp=None,
                       warmup: int = 3) -> tuple:
    """Convenience wrapper: builds the static input buffer and the capture
    closure for the common (model, loss_fn(logits, x), optimizer) loop.

    Returns ``(graphed, static_x)``; per batch do
    ``static_x.copy_(batch); graphed.replay()``.
    """
    static_x = example_input.clone()

    def step():
        if ddp is not None:
            ddp.zero_grad_buffers()
        else:
            optimizer.zero_grad(set_to_none=False)
        logits = model(static_x) if ddp is None else ddp(static_x)
        loss = loss_fn(logits, static_x)
        loss.backward()
        if ddp is not None:
            ddp.grad_sync()
        optimizer.step()
        return loss

    return GraphedStep(step, warmup=warmup), static_x
