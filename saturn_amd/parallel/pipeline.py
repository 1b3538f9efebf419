"""GPipe-style pipeline parallelism: single process, g devices, microbatched
fill-drain.

MI355X-native replacement for the reference's torchgpipe dependency
(``examples/wikitext103/executors/Pipeline.py:24-63``).  Stages are
consecutive slices of an ``nn.Sequential`` placed on successive devices;
activations hop stages with async device-to-device copies — on an 8xMI355X
node every GPU pair is one xGMI hop, so stage placement is
free and ``Tensor.to(device, non_blocking=True)`` rides the direct link.
Microbatches are enqueued host-side in pipeline order; HIP streams per
device overlap stage s of chunk j with stage s-1 of chunk j+1.
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import torch
import torch.nn as nn


def _greedy_partition(sizes: List[float], n_stages: int) -> List[int]:
    assert len(sizes) >= n_stages, "fewer layers than stages"
    balance: List[int] = []
    remaining = float(sum(sizes))
    stages_left = n_stages
    acc, cnt = 0.0, 0
    for i, s in enumerate(sizes):
        acc += s
        cnt += 1
        layers_after = len(sizes) - i - 1
        if stages_left > 1 and (
            acc >= remaining / stages_left or layers_after == stages_left - 1
        ):
            balance.append(cnt)
            remaining -= acc
            acc, cnt = 0.0, 0
            stages_left -= 1
    balance.append(cnt)
    return balance


def balance_by_params(seq: nn.Sequential, n_stages: int) -> List[int]:
    """Split layers into n_stages with ~equal parameter bytes (deterministic
    default; ``balance_by_time`` is the search() refinement)."""
    sizes = [
        sum(p.numel() * p.element_size() for p in m.parameters()) + 1.0
        for m in seq
    ]
    return _greedy_partition(sizes, n_stages)


def balance_by_time(
    seq: nn.Sequential,
    sample: torch.Tensor,
    n_stages: int,
    device: Optional[torch.device] = None,
    reps: int = 3,
) -> List[int]:
    """Split layers by measured per-layer forward time (the reference's
    torchgpipe ``balance_by_time``, Pipeline.py:94-103): wrong stage loads
    under parameter balance when layers are uneven — GPT-J's embedding and
    lm-head stages dwarf a block's bytes but not its time.

    Profiles on ``device`` (whole model must fit one device; the caller
    falls back to parameter balance on OOM).  The model is left on that
    device — callers re-place stages afterwards.
    """
    from timeit import default_timer as timer

    if device is None:
        device = next(
            (p.device for p in seq.parameters()), torch.device("cpu")
        )
    n = len(seq)
    times = [0.0] * n

    def sync() -> None:
        if device.type == "cuda":
            torch.cuda.synchronize(device)

    with torch.no_grad():
        for rep in range(reps + 1):  # rep 0 = warmup (also moves layers)
            h = sample.to(device)
            for i, m in enumerate(seq):
                if rep == 0:
                    m.to(device)
                sync()
                t0 = timer()
                h = m(h)
                sync()
                if rep > 0:
                    times[i] += timer() - t0
    return _greedy_partition([t + 1e-9 for t in times], n_stages)


class PipelinedModel(nn.Module):
    """An nn.Sequential split over devices with microbatch execution."""

    def __init__(
        self,
        seq: nn.Sequential,
        devices: Sequence[torch.device],
        balance: Optional[List[int]] = None,
        chunks: int = 4,
        checkpoint_activations: bool = False,
    ) -> None:
        super().__init__()
        self.devices = [torch.device(d) for d in devices]
        self.chunks = chunks
        self.checkpoint_activations = checkpoint_activations
        if balance is None:
            balance = balance_by_params(seq, len(self.devices))
        assert sum(balance) == len(seq), (balance, len(seq))
        self.balance = balance
        self.stages = nn.ModuleList()
        it = iter(seq)
        for n, dev in zip(balance, self.devices):
            layers = [next(it) for _ in range(n)]
            self.stages.append(nn.Sequential(*layers).to(dev))

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        """Microbatched forward; returns concatenated output on the last
        device.  Keep grad enabled for training — autograd tracks the
        cross-device copies."""
        parts = x.chunk(self.chunks, dim=0)
        outs = []
        for part in parts:
            h = part.to(self.devices[0], non_blocking=True)
            for stage, dev in zip(self.stages, self.devices):
                h = h.to(dev, non_blocking=True)
                if self.checkpoint_activations and torch.is_grad_enabled() and h.is_floating_point():
                    from torch.utils.checkpoint import checkpoint

                    h = checkpoint(stage, h, use_reentrant=False)
                else:
                    h = stage(h)
            outs.append(h)
        return torch.cat(outs, dim=0)

    def parameters_by_stage(self):
        return [list(s.parameters()) for s in self.stages]
