"""Flat-parameter model base.

FL operates on flat parameter vectors: delta extraction, candidate
reconstruction, FedAvg, and the RCCL all-gather all want ONE contiguous
buffer per model. So every model here owns a single
fp32 master buffer `flat`; layer weights are autograd views into it.
loss.backward() therefore accumulates the WHOLE gradient into
`flat.grad` — one fused SGD/Adam kernel updates the entire model, and
`(flat0 - flat) / lr` is one AXPY (reference main.py:153-154).
"""
from __future__ import annotations

import math
from typing import Dict, List, Optional, Sequence, Tuple

import torch

from bflc_amd.ops import functional as O


class FlatModel:
    """Base: subclasses define specs() and forward()."""

    def __init__(self, cfg, device: torch.device,
                 compute_dtype: Optional[torch.dtype] = None) -> None:
        self.cfg = cfg
        self.device = torch.device(device)
        if compute_dtype is None:
            compute_dtype = (torch.bfloat16 if self.device.type == "cuda"
                             and cfg.dtype == "bf16" else torch.float32)
        self.compute_dtype = compute_dtype

        self._specs: List[Tuple[str, Tuple[int, ...], str]] = list(self.specs())
        self._offsets: Dict[str, Tuple[int, Tuple[int, ...]]] = {}
        off = 0
        for name, shape, _ in self._specs:
            n = int(math.prod(shape))
            self._offsets[name] = (off, shape)
            off += n
        self.numel = off
        self.flat = torch.zeros(off, dtype=torch.float32, device=self.device)
        self.flat.requires_grad_(True)

    # -- subclass interface -------------------------------------------------
    def specs(self) -> Sequence[Tuple[str, Tuple[int, ...], str]]:
        """[(name, shape, init)] with init in {zeros, kaiming, xavier}."""
        raise NotImplementedError

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    # -- params -------------------------------------------------------------
    def init_params(self, seed: int) -> None:
        """Deterministic init from a CPU generator — identical on every
        rank, which is what makes the replicated global model consistent
        at epoch 0 (the reference starts from zeros, .h:30-34)."""
        gen = torch.Generator().manual_seed(seed)
        host = torch.zeros(self.numel, dtype=torch.float32)
        for name, shape, init in self._specs:
            off, _ = self._offsets[name]
            n = int(math.prod(shape))
            v = host[off:off + n].view(shape)
            if init == "zeros":
                pass
            elif init == "kaiming":
                fan_in = int(math.prod(shape[1:])) if len(shape) > 1 else shape[0]
                std = math.sqrt(2.0 / max(fan_in, 1))
                v.copy_(torch.randn(shape, generator=gen) * std)
            elif init == "xavier":
                fan_in = int(math.prod(shape[1:])) if len(shape) > 1 else shape[0]
                fan_out = shape[0]
                std = math.sqrt(2.0 / max(fan_in + fan_out, 1))
                v.copy_(torch.randn(shape, generator=gen) * std)
            else:
                raise ValueError(init)
        with torch.no_grad():
            self.flat.copy_(host.to(self.device))

    def p(self, name: str) -> torch.Tensor:
        """Autograd view of a parameter in compute dtype (fresh per call:
        views must be re-derived after in-place flat updates)."""
        off, shape = self._offsets[name]
        n = int(math.prod(shape))
        v = self.flat[off:off + n].view(shape)
        if self.compute_dtype != torch.float32:
            v = v.to(self.compute_dtype)
        return v

    def get_flat(self) -> torch.Tensor:
        return self.flat.detach().clone()

    def set_flat(self, v: torch.Tensor) -> None:
        with torch.no_grad():
            self.flat.copy_(v.to(self.device))

    def zero_grad(self) -> None:
        if self.flat.grad is not None:
            self.flat.grad.detach_()
            self.flat.grad.zero_()

    # -- train/eval ---------------------------------------------------------
    def loss(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        return O.softmax_cross_entropy(self.forward(self._cast(x)), y)

    def _cast(self, x: torch.Tensor) -> torch.Tensor:
        return x.to(device=self.device, dtype=self.compute_dtype)

    @torch.no_grad()
    def accuracy(self, x: torch.Tensor, y: torch.Tensor,
                 batch: int = 4096) -> float:
        """mean(argmax(pred)==y) over the set, batched (reference
        main.py:172-193 local_testing / 285-306 global_testing)."""
        total, correct = 0, 0.0
        for i in range(0, x.shape[0], batch):
            xb = self._cast(x[i:i + batch])
            yb = y[i:i + batch].to(self.device)
            logits = self.forward(xb)
            correct += O.accuracy(logits, yb) * xb.shape[0]
            total += xb.shape[0]
        return correct / max(total, 1)
