"""hipGraph-captured local-train step.

The FL local-train inner loop is a fixed kernel DAG: parameters are
per-parameter autograd leaves ALIASING one flat compute buffer at fixed
device addresses (models/base.py), the optimizer is one fused flat
kernel, and every minibatch has the same shape (the trainer drops the
ragged tail, engine.py). A profiled round is made of many 8-25 us
kernels (profiles/r01_kernel_stats_nhwc.md), so per-kernel launch gaps
are a real cost; capturing one

    zero-visible-state -> forward -> backward -> fused optimizer -> loss+=

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

SGD captures trivially. Adam captures through the graph-safe kernel
pair ``adam_tick`` + ``adam_master_dev`` (csrc/hip/elementwise.hip):
the step counter and the bias corrections (1-beta^t) live in DEVICE
buffers advanced inside the captured step, so nothing epoch-dependent
is baked in; ``reset_state()`` zeroes (m, v, step) between clients —
the reference builds a fresh optimizer per round (main.py:109,126).
Adam-on-GPU-fp32 (no bf16 shadow) stays eager: only the master+shadow
kernel has the device-step variant, and fp32 compute is a test-only
configuration.

Capture executes real kernels, so construction MUTATES the weights —
callers build the stepper before loading the round's global weights
(engine.py does this), or snapshot/restore around construction.

All captures in one engine share ONE graph memory pool (pass
``pool=graph.pool()`` of the first capture): per-scorer private pools
each held full-shard forward activations and grew GPU memory with the
number of local scorers (round-1 ADVICE item).
"""
from __future__ import annotations

import torch


class GraphedTrainStep:
    """One captured (forward, backward, fused-optimizer) step over
    static input buffers; ``step()`` copies a batch in and replays."""

    def __init__(self, model, lr: float, x_proto: torch.Tensor,
                 y_proto: torch.Tensor, optimizer: str = "sgd",
                 warmup: int = 2, pool=None) -> None:
        self.model = model
        self.optimizer = optimizer
        self.sx = torch.empty_like(x_proto)
        self.sy = torch.empty_like(y_proto)
        self.cost = torch.zeros((), device=model.device,
                                dtype=torch.float32)
        self.sx.copy_(x_proto)
        self.sy.copy_(y_proto)
        if optimizer == "adam":
            if model.cflat is model.flat:
                raise RuntimeError(
                    "graphed Adam needs the bf16-shadow model (the "
                    "device-step kernel is adam_master_graph_)")
            self.m = torch.zeros_like(model.flat)
            self.v = torch.zeros_like(model.flat)
            self.step_t = torch.zeros(1, device=model.device,
                                      dtype=torch.int32)
            self.bc = torch.zeros(2, device=model.device,
                                  dtype=torch.float32)
        elif optimizer != "sgd":
            raise ValueError(optimizer)
        self._lr = float(lr)

        # warmup on a side stream (materializes autograd engine state,
        # cat/workspace allocations) so none of it happens mid-capture
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(max(warmup, 1)):
                self._one_step()
        torch.cuda.current_stream().wait_stream(side)

        # grads=None at capture: backward's grad tensors are allocated
        # from the graph pool and rewritten in place on every replay
        self.model.zero_grad()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph, pool=pool):
            loss = self.model.loss(self.sx, self.sy)
            loss.backward()
            self._optim_step()
            self.cost.add_(loss.detach().float())

    def _optim_step(self) -> None:
        if self.optimizer == "adam":
            self.model.adam_step_graph(self.m, self.v, self.step_t,
                                       self.bc, self._lr)
        else:
            self.model.sgd_step(self._lr)

    def _one_step(self) -> None:
        self.model.zero_grad()
        loss = self.model.loss(self.sx, self.sy)
        loss.backward()
        self._optim_step()

    def reset_state(self) -> None:
        """Fresh optimizer for a new client/round (reference builds a
        new TF graph per round, main.py:109)."""
        if self.optimizer == "adam":
            self.m.zero_()
            self.v.zero_()
            self.step_t.zero_()

    def pool(self):
        return self.graph.pool()

    def matches(self, xb: torch.Tensor, yb: torch.Tensor) -> bool:
        return (xb.shape == self.sx.shape and xb.dtype == self.sx.dtype
                and yb.shape == self.sy.shape)

    def step(self, xb: torch.Tensor, yb: torch.Tensor) -> None:
        self.sx.copy_(xb)
        self.sy.copy_(yb)
        self.graph.replay()


class GraphedLocalTrain:
    """A client's ENTIRE local training pass as one graph: every
    minibatch step (forward, backward, fused optimizer, loss
    accumulate) of `local_epochs x total_batches` captured back to back
    with the client's OWN shard slices baked in (shard tensors are
    engine-lifetime persistent, so no per-batch input copies at all).
    One replay per client per round instead of one per batch — the
    per-round launch/copy overhead of the train phase collapses to a
    single dispatch."""

    def __init__(self, model, lr: float, shard_x: torch.Tensor,
                 shard_y: torch.Tensor, batch_size: int,
                 local_epochs: int, optimizer: str = "sgd",
                 pool=None) -> None:
        self.model = model
        self.optimizer = optimizer
        self.cost = torch.zeros((), device=model.device,
                                dtype=torch.float32)
        n = shard_x.shape[0]
        bs = min(batch_size, n)
        self.total_batches = max(n // bs, 1)
        self.n_steps = self.total_batches * local_epochs
        if optimizer == "adam":
            if model.cflat is model.flat:
                raise RuntimeError("graphed Adam needs the bf16 shadow")
            self.m = torch.zeros_like(model.flat)
            self.v = torch.zeros_like(model.flat)
            self.step_t = torch.zeros(1, device=model.device,
                                      dtype=torch.int32)
            self.bc = torch.zeros(2, device=model.device,
                                  dtype=torch.float32)
        elif optimizer != "sgd":
            raise ValueError(optimizer)
        self._lr = float(lr)
        self._batches = [(shard_x[b * bs:(b + 1) * bs],
                          shard_y[b * bs:(b + 1) * bs])
                         for b in range(self.total_batches)] * local_epochs

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):
                self._one_step(*self._batches[0])
        torch.cuda.current_stream().wait_stream(side)
        self.model.zero_grad()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph, pool=pool):
            for xb, yb in self._batches:
                self._one_step(xb, yb, accumulate=True)

    def _optim_step(self) -> None:
        if self.optimizer == "adam":
            self.model.adam_step_graph(self.m, self.v, self.step_t,
                                       self.bc, self._lr)
        else:
            self.model.sgd_step(self._lr)

    def _one_step(self, xb, yb, accumulate: bool = False) -> None:
        self.model.zero_grad()
        loss = self.model.loss(xb, yb)
        loss.backward()
        self._optim_step()
        if accumulate:
            self.cost.add_(loss.detach().float())

    def pool(self):
        return self.graph.pool()

    def run(self) -> torch.Tensor:
        """One replay = the whole local training pass. Caller loads the
        round's global weights first (set_flat). Returns the summed
        per-batch loss (device scalar)."""
        self.cost.zero_()
        if self.optimizer == "adam":
            self.m.zero_()
            self.v.zero_()
            self.step_t.zero_()
        self.graph.replay()
        return self.cost


class GraphedScorePhase:
    """One scorer's ENTIRE committee-scoring pass as one graph: for
    each of the K quota slots, build candidate = global - lr*stack[k]
    in place, load it, and evaluate accuracy on the scorer's shard —
    K candidate evaluations and their accuracy reads in a single
    replay. `cand_stack` must be the engine's persistent [K, P] packed
    buffer (same device pointers every round)."""

    def __init__(self, model, shard_x: torch.Tensor,
                 shard_y: torch.Tensor, cand_stack: torch.Tensor,
                 global_flat: torch.Tensor, lr: float, pool=None) -> None:
        from bflc_amd.ops import functional as O
        self.model = model
        self.K = cand_stack.shape[0]
        self.accs = torch.zeros(self.K, device=model.device,
                                dtype=torch.float32)
        lr = float(lr)

        # candidate load is ONE fused kernel writing the compute-dtype
        # shadow directly (the eval forward's only weight consumer);
        # the fp32 master stays stale during scoring — every later
        # phase reloads it via set_flat (ops/functional.py score_load_)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            O.score_load_(model.cflat.data, global_flat, cand_stack[0], lr)
            model.accuracy_t(shard_x, shard_y)
        torch.cuda.current_stream().wait_stream(side)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph, pool=pool):
            for k in range(self.K):
                O.score_load_(model.cflat.data, global_flat,
                              cand_stack[k], lr)
                self.accs[k] = model.accuracy_t(shard_x, shard_y)

    def pool(self):
        return self.graph.pool()

    def run(self) -> torch.Tensor:
        """Replay; returns a snapshot of the K accuracies (device)."""
        self.graph.replay()
        return self.accs.clone()

    def run_inplace(self) -> torch.Tensor:
        """Replay; returns the live accs buffer (no allocation) — for
        concurrent per-scorer stream replays, where the caller reads
        the buffer only after joining all scorer streams."""
        self.graph.replay()
        return self.accs


class GraphedScore:
    """Captured committee-scoring evaluation: load a candidate flat
    vector into the model (copy + shadow refresh) and run accuracy over
    the scorer's own shard — the shard tensors are baked into the graph
    (they never change), only the candidate is copied in per replay.
    The per-round scoring cost scales with the number of admitted
    candidates (committee size x quota at 8 nodes), so this is the
    phase that grows with world size."""

    def __init__(self, model, shard_x: torch.Tensor,
                 shard_y: torch.Tensor, pool=None) -> None:
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
        with torch.cuda.graph(self.graph, pool=pool):
            model.set_flat(self.cand)
            self.acc = model.accuracy_t(shard_x, shard_y)

    def pool(self):
        return self.graph.pool()

    def score(self, cand: torch.Tensor) -> torch.Tensor:
        """Returns a device scalar snapshot (no host sync)."""
        self.cand.copy_(cand)
        self.graph.replay()
        return self.acc.clone()

    def score_candidate(self, global_flat: torch.Tensor, lr: float,
                        delta: torch.Tensor) -> torch.Tensor:
        """Build candidate = global - lr*delta DIRECTLY in the graph's
        input buffer (reference main.py:215-216) — no per-candidate
        clone+copy round trip (the engine's scoring loop runs
        committee x quota of these per round)."""
        from bflc_amd.ops import functional as O
        self.cand.copy_(global_flat)
        O.axpy_(self.cand, -lr, delta)
        self.graph.replay()
        return self.acc.clone()
