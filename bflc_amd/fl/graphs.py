"""hipGraph-captured local-train step.

The FL local-train inner loop is a fixed kernel DAG: parameters are
per-parameter autograd leaves ALIASING one flat compute buffer at fixed
device addresses (models/base.py), the optimizer is one fused flat
kernel, and every minibatch has the same shape (the trainer drops the
ragged tail, engine.py). A profiled round is made of many 8-25 us
kernels (profiles/r01_kernel_stats_nhwc.md), so per-kernel launch gaps
are a real cost; capturing one

    zero-visible-state -> forward -> backward -> fused SGD -> loss+=

step into a hipGraph and replaying it once per batch launches the whole
step as a single unit.

Capture safety relies on invariants the rest of the framework already
guarantees:
  * ``model.flat`` / ``model.cflat`` never reallocate — ``set_flat``
    copies INTO them — so kernel pointers baked at capture stay valid
    across rounds and clients;
  * the train step performs no host sync (the loss accumulates into a
    device scalar; ``accuracy_t`` is device-resident; the only
    ``.item()`` calls in the stack are in FedAvg, outside this region);
  * intermediate/grad tensors allocated during capture live in the
    graph's private memory pool and are rewritten in place on replay.

SGD only: Adam's per-step bias correction is a host scalar that would
be baked in at capture time; the eager path keeps handling Adam.
Capture executes real kernels, so construction MUTATES the weights —
callers build the stepper before loading the round's global weights
(engine.py does this), or snapshot/restore around construction.
"""
from __future__ import annotations

import torch


class GraphedTrainStep:
    """One captured (forward, backward, fused-SGD) step over static
    input buffers; ``step()`` copies a batch in and replays."""

    def __init__(self, model, lr: float, x_proto: torch.Tensor,
                 y_proto: torch.Tensor, warmup: int = 2) -> None:
        self.model = model
        self.sx = torch.empty_like(x_proto)
        self.sy = torch.empty_like(y_proto)
        self.cost = torch.zeros((), device=model.device,
                                dtype=torch.float32)
        self.sx.copy_(x_proto)
        self.sy.copy_(y_proto)

        # warmup on a side stream (materializes autograd engine state,
        # cat/workspace allocations) so none of it happens mid-capture
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(max(warmup, 1)):
                self._eager_step(lr)
        torch.cuda.current_stream().wait_stream(side)

        # grads=None at capture: backward's grad tensors are allocated
        # from the graph pool and rewritten in place on every replay
        self.model.zero_grad()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            loss = self.model.loss(self.sx, self.sy)
            loss.backward()
            self.model.sgd_step(lr)
            self.cost.add_(loss.detach().float())

    def _eager_step(self, lr: float) -> None:
        self.model.zero_grad()
        loss = self.model.loss(self.sx, self.sy)
        loss.backward()
        self.model.sgd_step(lr)

    def matches(self, xb: torch.Tensor, yb: torch.Tensor) -> bool:
        return (xb.shape == self.sx.shape and xb.dtype == self.sx.dtype
                and yb.shape == self.sy.shape)

    def step(self, xb: torch.Tensor, yb: torch.Tensor) -> None:
        self.sx.copy_(xb)
        self.sy.copy_(yb)
        self.graph.replay()


class GraphedScore:
    """Captured committee-scoring evaluation: load a candidate flat
    vector into the model (copy + shadow refresh) and run accuracy over
    the scorer's own shard — the shard tensors are baked into the graph
    (they never change), only the candidate is copied in per replay.
    The per-round scoring cost scales with the number of admitted
    candidates (committee size x quota at 8 nodes), so this is the
    phase that grows with world size."""

    def __init__(self, model, shard_x: torch.Tensor,
                 shard_y: torch.Tensor) -> None:
        self.model = model
        self.cand = torch.empty_like(model.flat)
        self.cand.copy_(model.flat.detach())
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            model.set_flat(self.cand)
            model.accuracy_t(shard_x, shard_y)
        torch.cuda.current_stream().wait_stream(side)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            model.set_flat(self.cand)
            self.acc = model.accuracy_t(shard_x, shard_y)

    def score(self, cand: torch.Tensor) -> torch.Tensor:
        """Returns a device scalar snapshot (no host sync)."""
        self.cand.copy_(cand)
        self.graph.replay()
        return self.acc.clone()
