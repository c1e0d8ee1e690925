"""Flat-parameter model base.

FL operates on flat parameter vectors: delta extraction, candidate
reconstruction, FedAvg, and the RCCL all-gather all want ONE contiguous
buffer per model. So every model here owns a single
fp32 master buffer `flat`; layer weights are per-parameter autograd
LEAVES whose storage aliases the flat compute buffer (parameter-sized
grads — a single flat leaf made autograd materialize full-model-sized
slice-backward grads per parameter). grad_flat() cats the per-parameter
grads into one contiguous vector, one fused SGD/Adam kernel updates the
entire model, and `(flat0 - flat) / lr` is one AXPY (reference
main.py:153-154).
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
        # fp32 master vector; when computing in bf16 a separate bf16
        # shadow leaf carries the forward/backward (its .grad is the flat
        # bf16 gradient) and the fused sgd_master_/adam_master_ kernels
        # update master + shadow in one pass — no per-parameter casts.
        self.flat = torch.zeros(off, dtype=torch.float32, device=self.device)
        if self.compute_dtype == torch.float32:
            self.cflat = self.flat
        else:
            self.cflat = torch.zeros(off, dtype=self.compute_dtype,
                                     device=self.device)
        # Per-parameter autograd LEAVES aliasing the flat compute buffer
        # (set_ shares storage). With ONE flat leaf, every parameter's
        # slice-backward materialized a full-numel zeros + add per
        # parameter per backward (~160 x 25M-element passes on
        # ResNet-50, ~20% of a round); per-parameter leaves make each
        # grad parameter-sized, and grad_flat() cats them back into one
        # contiguous vector matching the offset layout.
        self._params: Dict[str, torch.Tensor] = {}
        for name, shape, _ in self._specs:
            o, _ = self._offsets[name]
            t = torch.empty(0, dtype=self.compute_dtype, device=self.device)
            t.set_(self.cflat, o, shape)
            t.requires_grad_(True)
            self._params[name] = t

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
        self._sync_shadow()

    def _sync_shadow(self) -> None:
        if self.cflat is self.flat:
            return
        with torch.no_grad():
            if self.device.type == "cuda":
                from bflc_amd.ops.functional import hip_ops
                hip_ops().refresh_shadow_(self.flat.detach(),
                                          self.cflat.data)
            else:
                self.cflat.data.copy_(self.flat.detach())

    def p(self, name: str) -> torch.Tensor:
        """Leaf parameter tensor (compute dtype) aliasing the flat
        buffer; in-place flat updates are visible through it."""
        return self._params[name]

    def get_flat(self) -> torch.Tensor:
        return self.flat.detach().clone()

    def set_flat(self, v: torch.Tensor) -> None:
        with torch.no_grad():
            self.flat.copy_(v.to(self.device))
        self._sync_shadow()

    def zero_grad(self) -> None:
        for t in self._params.values():
            t.grad = None

    def grad_flat(self) -> torch.Tensor:
        """All parameter grads as ONE contiguous vector (spec order ==
        offset order), built with a single cat kernel."""
        gs = []
        for name, shape, _ in self._specs:
            g = self._params[name].grad
            assert g is not None, f"no grad for {name}"
            gs.append(g.reshape(-1))
        return torch.cat(gs)

    # -- fused optimizer steps (fp32 master + shadow refresh) ----------
    def sgd_step(self, lr: float) -> None:
        g = self.grad_flat()
        if self.cflat is self.flat:
            from bflc_amd.ops.functional import sgd_step_
            sgd_step_(self.flat.data, g, lr)
        elif self.device.type == "cuda":
            from bflc_amd.ops.functional import hip_ops
            hip_ops().sgd_master_(self.flat.detach(), self.cflat.data, g,
                                  float(lr))
        else:  # CPU oracle of the fused master-SGD kernel
            with torch.no_grad():
                self.flat.add_(g.float(), alpha=-lr)
                self.cflat.data.copy_(self.flat.to(self.compute_dtype))

    def adam_step(self, m: torch.Tensor, v: torch.Tensor, step: int,
                  lr: float) -> None:
        g = self.grad_flat()
        if self.cflat is self.flat:
            from bflc_amd.ops.functional import adam_step_
            adam_step_(self.flat.data, g, m, v, step, lr)
        elif self.device.type == "cuda":
            from bflc_amd.ops.functional import hip_ops
            hip_ops().adam_master_(self.flat.detach(), self.cflat.data, g,
                                   m, v, int(step), float(lr), 0.9, 0.999,
                                   1e-8)
        else:  # CPU oracle of the fused master-Adam kernel
            from bflc_amd.ops.functional import adam_step_
            with torch.no_grad():
                adam_step_(self.flat.data, g.float(), m, v, step, lr)
                self.cflat.data.copy_(self.flat.to(self.compute_dtype))

    def adam_step_graph(self, m: torch.Tensor, v: torch.Tensor,
                        step_t: torch.Tensor, bc: torch.Tensor,
                        lr: float) -> None:
        """hipGraph-capturable Adam: the step counter and bias
        corrections live in device buffers advanced by the fused kernel
        pair (csrc/hip/elementwise.hip adam_tick/adam_master_dev) — so
        the whole step captures without baking in a host-side epoch."""
        assert self.cflat is not self.flat, "needs the bf16 shadow model"
        g = self.grad_flat()
        from bflc_amd.ops.functional import hip_ops
        hip_ops().adam_master_graph_(self.flat.detach(), self.cflat.data,
                                     g, m, v, step_t, bc, float(lr), 0.9,
                                     0.999, 1e-8)

    # -- train/eval ---------------------------------------------------------
    def loss(self, x: torch.Tensor, y: torch.Tensor) -> torch.Tensor:
        return O.softmax_cross_entropy(self.forward(self._cast(x)), y)

    def _cast(self, x: torch.Tensor) -> torch.Tensor:
        return x.to(device=self.device, dtype=self.compute_dtype)

    @torch.no_grad()
    def accuracy_t(self, x: torch.Tensor, y: torch.Tensor,
                   batch: int = 4096) -> torch.Tensor:
        """Device-resident accuracy (no host sync): callers batching
        several evaluations sync once at the end."""
        total = 0
        correct = torch.zeros((), device=self.device)
        for i in range(0, x.shape[0], batch):
            xb = self._cast(x[i:i + batch])
            yb = y[i:i + batch].to(self.device)
            logits = self.forward(xb)
            correct += O.accuracy_t(logits, yb) * xb.shape[0]
            total += xb.shape[0]
        return correct / max(total, 1)

    @torch.no_grad()
    def accuracy(self, x: torch.Tensor, y: torch.Tensor,
                 batch: int = 4096) -> float:
        """mean(argmax(pred)==y) over the set, batched (reference
        main.py:172-193 local_testing / 285-306 global_testing)."""
        return float(self.accuracy_t(x, y, batch))
