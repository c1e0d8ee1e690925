"""Barrier-driven committee-consensus FL round engine.

One process per GPU; each rank owns a contiguous slice of the FL
clients and a full replica of the deterministic ledger. A round replays
the reference protocol without any polling or sleeps:

  phase T  designated trainers run the local HIP train step
  phase U  one rank-ordered all-gather publishes (meta, delta) updates;
           every rank feeds ALL submissions to its ledger replica in the
           same order => identical admission decisions
           (reference UploadLocalUpdate guards, .cpp:215-257)
  phase S  committee clients score every accepted candidate on their own
           shard (reference local_scoring, main.py:196-217)
  phase V  one all-gather publishes score maps; the comm_count-th feed
           triggers the aggregation decision (.cpp:259-297)
  phase A  every rank applies the identical weighted-FedAvg kernel with
           a fixed accumulation order and commits (.cpp:349-455)

The reference resolves the 16-trainers-race-for-10-slots with
first-come PBFT ordering; here admission order is the deterministic
(rank, client) order, so ranks can PREDICT which updates would be
admitted and skip training the surplus trainers entirely — same
semantics (rejected updates never influence state), less wasted compute.

Execution model (GPU): each local client's ENTIRE training pass and
ENTIRE candidate-scoring sweep are captured once as hipGraphs
(fl/graphs.py) against that client's own model replica and private
graph pool; within a phase, the participants' replays run CONCURRENTLY
on slot-indexed HIP streams and join at the phase barrier. Per-client
kernel order is unchanged, so results are bitwise-identical to the
sequential path (BFLC_STREAMS=0). The FL-model kernels are
issue/launch-bound rather than HBM-bound (profiles/r02_pmc_*.md), which
is why overlap — not more per-kernel bandwidth — is what shortens the
round.

Data plane (flat fp32 delta tensors, the global model) stays resident on
the GPU; the ledger's blobs hold b"" markers in engine mode (the
ABI-compatible JSON blob path lives in bflc_amd.chain.client).
"""
from __future__ import annotations

import os
import time
import warnings
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import torch

from bflc_amd._ledger import Admit, CommitteeLedger
from bflc_amd.chain.identity import (KeyTable, scores_payload,
                                     update_payload)
from bflc_amd.comm import Transport
from bflc_amd.config import FLConfig
from bflc_amd.data.synthetic import Shard
from bflc_amd.models import build_model
from bflc_amd.ops import functional as O


class StragglerError(RuntimeError):
    """A collective phase failed (dead or too-slow peer). The engine
    aborts the round cleanly after writing a rank-local checkpoint; the
    reference survived crashed trainers by quota over-provisioning
    (CommitteePrecompiled.h:15, only 10 of 16 need to upload) — a
    barrier-driven engine turns the hang into this diagnosable error
    instead, bounded by Transport.timeout_s."""


def client_rank(client_idx: int, n_clients: int, world: int) -> int:
    """Contiguous array_split-style client->rank assignment."""
    base, rem = divmod(n_clients, world)
    # ranks 0..rem-1 own (base+1) clients
    cut = rem * (base + 1)
    if client_idx < cut:
        return client_idx // (base + 1)
    return rem + (client_idx - cut) // max(base, 1)


@dataclass
class RoundStats:
    epoch: int
    wall_s: float
    train_s: float
    gather_s: float
    score_s: float
    aggregate_s: float
    global_loss: float
    n_updates: int
    n_selected: int
    samples_trained: int
    test_acc: Optional[float] = None


class FLEngine:
    def __init__(self, cfg: FLConfig, transport: Transport,
                 shards: List[Shard], test_shard: Optional[Shard] = None,
                 metrics_path: Optional[str] = None) -> None:
        from bflc_amd.utils import JsonlLogger
        self.cfg = cfg
        self.t = transport
        self.device = transport.device
        self.rank = transport.rank
        self.world = transport.world_size

        # client ownership
        self.origins = [f"node_{i}" for i in range(cfg.client_num)]
        self.local_clients = [i for i in range(cfg.client_num)
                              if client_rank(i, cfg.client_num, self.world)
                              == self.rank]
        self.shards = {i: shards[i].to(self.device) for i in self.local_clients}
        self.test_shard = test_shard.to(self.device) if test_shard else None

        # one shared model instance per rank (set_flat per client)
        self.model = build_model(cfg, self.device)
        self.global_flat = self.model.get_flat()  # device-resident fp32

        # per-origin HMAC keys (reference get_batch_accounts.sh: one
        # identity per client); every gathered submission is verified
        # before it reaches the ledger (chain/identity.py)
        self.keys = KeyTable(self.origins, cfg.seed)

        # replicated ledger: registration = deterministic replay of the
        # rank-ordered client lists (reference RegisterNode, .cpp:168-190)
        self.ledger = CommitteeLedger(cfg.ledger_config())
        gathered = self.t.all_gather_objects(self.local_clients)
        for rank_clients in gathered:
            for i in rank_clients:
                self.ledger.register_node(self.origins[i])
        assert self.ledger.epoch == 0, "registration did not start FL"
        self.ledger.set_global_model(b"")

        self._round = 0
        self.metrics = JsonlLogger(metrics_path, rank=self.rank)

        # hipGraph-captured train step (fl/graphs.py): replay one
        # captured kernel DAG per minibatch instead of relaunching
        # ~dozens of kernels. SGD-only; falls back to eager on CPU, on
        # Adam, or if capture fails on this ROCm build.
        self._steppers: Dict[tuple, object] = {}
        self._scorers: Dict[int, object] = {}
        # whole-phase graphs: one per client (full local training pass)
        # and one per scorer (all quota candidates in one replay) —
        # launch/copy overhead per round collapses to one dispatch per
        # phase participant (fl/graphs.py)
        self._train_graphs: Dict[int, object] = {}
        self._score_graphs: Dict[int, object] = {}
        self._cand_stack: Optional[torch.Tensor] = None
        self._graph_pool = None  # shared across all captures (one pool)
        # concurrent client streams: the FL-model kernels are
        # issue/launch-bound, not HBM-bound (profiles/r02_pmc_femnist.md)
        # — so a rank hosting several clients overlaps their whole-phase
        # graph replays on concurrent HIP streams (slot-indexed per
        # phase participant, see _client_stream). That requires one
        # model replica and one private graph pool per local client
        # (concurrently-replaying graphs must not share weight buffers
        # or pool memory); the cost is #local_clients x (model +
        # activation footprint), sized for 288 GB HBM3E (ResNet-50 at 8
        # clients/rank ~ tens of GB). Per-client compute is unchanged
        # and each client's kernels stay serialized on its own stream,
        # so results are identical to the sequential path.
        self._client_models: Dict[int, object] = {}
        self._client_streams: Dict[int, object] = {}
        self._client_pools: Dict[int, object] = {}
        self._client_delta: Dict[int, torch.Tensor] = {}
        # round-persistent scratch (avoid per-round alloc/fill of
        # O(model)-sized tensors: the gather stack is ~400 MB/round on
        # ResNet-50)
        self._cand_buf: Optional[torch.Tensor] = None
        self._stack_buf: Optional[torch.Tensor] = None
        self._fedavg_buf: Optional[torch.Tensor] = None
        self._use_graphs = (cfg.use_graphs and self.device.type == "cuda"
                            and os.environ.get("BFLC_GRAPHS", "1") != "0")
        self._concurrent = (self._use_graphs
                            and os.environ.get("BFLC_STREAMS", "1") != "0")
        self._phase_debug = os.environ.get("BFLC_PHASE_DEBUG", "0") == "1"

    # ------------------------------------------------------------------
    def _planned_submitters(self) -> List[int]:
        """The deterministic prefix of trainers whose updates will be
        admitted (first needed_update_count in (rank, client) order)."""
        roles = self.ledger.roles()
        order: List[int] = []
        gathered_order = range(self.cfg.client_num)  # origins are id-sorted
        # global (rank, client-index) order == client index order because
        # assignment is contiguous ascending
        for i in gathered_order:
            role = roles.get(self.origins[i], "trainer")
            if role == "trainer" or self.cfg.self_scoring:
                order.append(i)
        return order[: self.cfg.needed_update_count]

    # ------------------------------------------------------------------
    def _graphed_stepper(self, shard: Shard, bs: int):
        """Lazily capture the train-step hipGraph (fl/graphs.py), one
        stepper per batch shape (heterogeneous shard sizes each get
        their own capture instead of silently running eager — round-1
        ADVICE item). All captures share one graph memory pool. Returns
        None — and stops trying — if capture is unavailable."""
        xb, yb = shard.x[:bs], shard.y[:bs]
        key = (tuple(xb.shape), xb.dtype, tuple(yb.shape))
        if key in self._steppers:
            return self._steppers[key]
        try:
            from bflc_amd.fl.graphs import GraphedTrainStep
            st = GraphedTrainStep(self.model, self.cfg.learning_rate, xb,
                                  yb, optimizer=self.cfg.optimizer,
                                  pool=self._graph_pool)
            if self._graph_pool is None:
                self._graph_pool = st.pool()
            self._steppers[key] = st
        except Exception as e:  # capture unsupported: eager fallback
            warnings.warn(f"hipGraph capture failed, running eager: {e}")
            self._use_graphs = False
            return None
        return self._steppers[key]

    # ------------------------------------------------------------------
    def _client_model(self, client: int):
        """Per-client model replica (concurrent mode): each local
        client's graphs capture against their OWN flat/cflat buffers so
        their replays can overlap on per-client streams."""
        if client not in self._client_models:
            self._client_models[client] = build_model(self.cfg, self.device)
        return self._client_models[client]

    def _client_stream(self, slot: int):
        """Stream for the slot-th CONCURRENT participant of the current
        phase (NOT per client): ROCm multiplexes streams onto
        GPU_MAX_HW_QUEUES (default 4) hardware queues and streams
        sharing a queue serialize — with one stream per CLIENT (8
        streams, roles rotating), a phase's 4 participants could land
        2-per-queue and run pairwise-serial (per-client stream timings
        showed exactly that). At most max-participants-per-phase
        streams ever exist, so each phase's replays map to distinct
        queues. Safe because phases are joined before the next launch
        and a client appears at most once per phase."""
        if slot not in self._client_streams:
            self._client_streams[slot] = torch.cuda.Stream()
        return self._client_streams[slot]

    def _capture_pool(self, client: int):
        """Graph memory pool for this client's captures. Concurrent
        mode: one PRIVATE pool per client (graphs replaying in parallel
        must not share pool memory); sequential mode: the one shared
        pool (round-1 ADVICE — serialized replays can share)."""
        if self._concurrent:
            return self._client_pools.get(client)
        return self._graph_pool

    def _note_pool(self, client: int, g) -> None:
        if self._concurrent:
            if self._client_pools.get(client) is None:
                self._client_pools[client] = g.pool()
        elif self._graph_pool is None:
            self._graph_pool = g.pool()

    # ------------------------------------------------------------------
    def _whole_train_graph(self, client: int):
        """Per-client whole-local-training graph (shard slices baked in,
        zero per-batch copies); None if capture is unavailable. In
        concurrent mode the graph is captured against the client's own
        model replica (g.model) so replays can overlap."""
        if client in self._train_graphs:
            return self._train_graphs[client]
        try:
            from bflc_amd.fl.graphs import GraphedLocalTrain
            shard = self.shards[client]
            model = (self._client_model(client) if self._concurrent
                     else self.model)
            g = GraphedLocalTrain(model, self.cfg.learning_rate,
                                  shard.x, shard.y, self.cfg.batch_size,
                                  self.cfg.local_epochs,
                                  optimizer=self.cfg.optimizer,
                                  pool=self._capture_pool(client))
            self._note_pool(client, g)
        except Exception as e:
            warnings.warn(f"whole-train graph capture failed for client "
                          f"{client}, using per-batch path: {e}")
            g = None
        self._train_graphs[client] = g
        return g

    def _local_train(self, client: int) -> Tuple[torch.Tensor, int, float]:
        """Local train step (reference local_training, main.py:103-158):
        start from the global model, run local_epochs passes of
        minibatch SGD/Adam, return pseudo-gradient delta=(W0-W)/lr."""
        cfg = self.cfg
        shard = self.shards[client]
        n = shard.n
        # whole-pass graph: capture BEFORE loading the round's weights
        # (capture executes real steps and mutates flat/cflat)
        wg = self._whole_train_graph(client) if self._use_graphs else None
        if wg is not None:
            wg.model.set_flat(self.global_flat)
            cost = wg.run()
            avg_cost = float(cost) / wg.n_steps
            delta = torch.empty_like(self.global_flat)
            O.delta_extract_(delta, self.global_flat, wg.model.flat.data,
                             cfg.learning_rate)
            return delta, n, avg_cost
        bs = min(cfg.batch_size, n)
        total_batches = max(n // bs, 1)
        # capture BEFORE loading the round's weights: graph warmup and
        # capture execute real SGD steps and mutate flat/cflat
        stepper = self._graphed_stepper(shard, bs) if self._use_graphs \
            else None
        self.model.set_flat(self.global_flat)
        if cfg.optimizer == "adam" and stepper is None:
            m = torch.zeros_like(self.global_flat)
            v = torch.zeros_like(self.global_flat)
            step = 0
        if stepper is not None:
            stepper.cost.zero_()
            stepper.reset_state()  # fresh Adam (m, v, step) per client
            for _ in range(cfg.local_epochs):
                for bi in range(total_batches):
                    stepper.step(shard.x[bi * bs:(bi + 1) * bs],
                                 shard.y[bi * bs:(bi + 1) * bs])
            cost_accum = stepper.cost
        else:
            # accumulate the loss on-device; ONE host sync per client
            # per round
            cost_accum = torch.zeros((), device=self.device)
            for _ in range(cfg.local_epochs):
                for bi in range(total_batches):
                    xb = shard.x[bi * bs:(bi + 1) * bs]
                    yb = shard.y[bi * bs:(bi + 1) * bs]
                    self.model.zero_grad()
                    loss = self.model.loss(xb, yb)
                    loss.backward()
                    if cfg.optimizer == "adam":
                        step += 1
                        self.model.adam_step(m, v, step, cfg.learning_rate)
                    else:
                        self.model.sgd_step(cfg.learning_rate)
                    cost_accum += loss.detach()
        avg_cost = float(cost_accum) / (total_batches * cfg.local_epochs)
        # delta = (W0 - W)/lr  (reference main.py:153-154)
        delta = torch.empty_like(self.global_flat)
        O.delta_extract_(delta, self.global_flat, self.model.flat.data,
                         cfg.learning_rate)
        return delta, n, avg_cost

    # ------------------------------------------------------------------
    def _pack_candidates(self, updates) -> bool:
        """Pack the admitted deltas into the persistent [quota, P]
        buffer the whole-scoring graphs bake in; False when the shape
        doesn't fit (degraded round) => per-candidate fallback."""
        K = len(updates)
        if not self._use_graphs or K != self.cfg.needed_update_count \
                or K == 0 or self.device.type != "cuda":
            return False
        if self._cand_stack is None:
            self._cand_stack = torch.empty(
                K, self.global_flat.numel(), dtype=torch.float32,
                device=self.device)
        for k, (_, delta) in enumerate(updates):
            self._cand_stack[k].copy_(delta)
        return True

    def _phase_scorer(self, scorer: int):
        """Per-scorer whole-phase scoring graph (all quota candidates in
        one replay); None if capture is unavailable."""
        if scorer in self._score_graphs:
            return self._score_graphs[scorer]
        try:
            from bflc_amd.fl.graphs import GraphedScorePhase
            shard = self.shards[scorer]
            model = (self._client_model(scorer) if self._concurrent
                     else self.model)
            g = GraphedScorePhase(model, shard.x, shard.y,
                                  self._cand_stack, self.global_flat,
                                  self.cfg.learning_rate,
                                  pool=self._capture_pool(scorer))
            self._note_pool(scorer, g)
        except Exception as e:
            warnings.warn(f"whole-scoring graph capture failed for "
                          f"scorer {scorer}, per-candidate path: {e}")
            g = None
        self._score_graphs[scorer] = g
        return g

    def _score_candidates(self, scorer: int,
                          updates: List[Tuple[str, torch.Tensor]],
                          packed: bool = False) -> Dict[str, float]:
        """Committee scoring (reference local_scoring, main.py:196-217):
        candidate = W0 - lr*delta, scored by accuracy on the scorer's own
        local shard."""
        if packed:
            g = self._phase_scorer(scorer)
            if g is not None:
                accs = g.run().tolist()  # one host sync for all K
                return {origin: accs[k]
                        for k, (origin, _) in enumerate(updates)}
        shard = self.shards[scorer]
        gs = self._graphed_scorer(scorer) if self._use_graphs else None
        lr = self.cfg.learning_rate
        accs: List[Tuple[str, torch.Tensor]] = []
        if gs is not None:
            for origin, delta in updates:
                # candidate built in the graph's own input buffer — no
                # per-candidate clone
                accs.append((origin,
                             gs.score_candidate(self.global_flat, lr,
                                                delta)))
        else:
            if self._cand_buf is None:
                self._cand_buf = torch.empty_like(self.global_flat)
            for origin, delta in updates:
                self._cand_buf.copy_(self.global_flat)
                O.axpy_(self._cand_buf, -lr, delta)
                self.model.set_flat(self._cand_buf)
                # device-resident: ONE host sync per scorer (below),
                # not one per candidate
                accs.append((origin,
                             self.model.accuracy_t(shard.x, shard.y)))
        return {origin: float(a) for origin, a in accs}

    def _graphed_scorer(self, scorer: int):
        """Per-scorer captured scoring graph (fl/graphs.py); capture
        mutates the scratch model, which every phase reloads anyway."""
        if scorer in self._scorers:
            return self._scorers[scorer]
        try:
            from bflc_amd.fl.graphs import GraphedScore
            shard = self.shards[scorer]
            gs = GraphedScore(self.model, shard.x, shard.y,
                              pool=self._graph_pool)
            if self._graph_pool is None:
                self._graph_pool = gs.pool()
            self._scorers[scorer] = gs
        except Exception as e:
            warnings.warn(f"score-graph capture failed, running eager: {e}")
            self._scorers[scorer] = None
        return self._scorers[scorer]

    # ------------------------------------------------------------------
    def _gather(self, phase: str, fn, *args):
        """Run one collective phase; a dead or timed-out peer becomes a
        clean StragglerError after a rank-local checkpoint (path:
        $BFLC_ABORT_CHECKPOINT, default under the system temp dir)."""
        try:
            return fn(*args)
        except StragglerError:
            raise
        except Exception as e:
            import tempfile
            path = os.environ.get(
                "BFLC_ABORT_CHECKPOINT",
                os.path.join(tempfile.gettempdir(),
                             f"bflc_abort_rank{self.rank}.pt"))
            note = ""
            try:
                self.save(path)
                note = f"; state checkpointed to {path}"
            except Exception:
                note = "; checkpoint write also failed"
            raise StragglerError(
                f"collective phase '{phase}' failed at epoch "
                f"{self.ledger.epoch} on rank {self.rank}/{self.world} "
                f"(dead or >={self.t.timeout_s:.0f}s-slow peer?): "
                f"{e}{note}") from e

    # ------------------------------------------------------------------
    def run_round(self, eval_global: bool = False) -> RoundStats:
        cfg, led = self.cfg, self.ledger
        if led.finished:
            raise RuntimeError(
                f"FL run finished: epoch {led.epoch} > max_epoch "
                f"{cfg.max_epoch} (reference clients exit here, "
                "main.py:251-252)")
        epoch = led.epoch
        t0 = time.perf_counter()

        # ---- phase T: local training on designated submitters ----------
        # Concurrent path: each local trainer's whole-training graph
        # replays on its own HIP stream (its own model replica and
        # private pool) — set_flat, the replay, and the fused delta
        # extraction are all enqueued per stream, joined once at the
        # phase end. Falls back to the sequential loop if any capture is
        # unavailable.
        submitters = self._planned_submitters()
        local_subs = [i for i in submitters if i in self.shards]
        local_updates: List[Tuple[str, torch.Tensor, int, float]] = []
        samples_trained = 0
        conc = self._concurrent and len(local_subs) > 1
        tgraphs = {}
        if self._concurrent and self._round == 0:
            # Pre-capture EVERY local client's training graph now:
            # committee rotation means each client eventually trains,
            # and a lazy capture during a later (timed/steady-state)
            # round would put one-time capture cost inside it.
            for i in self.local_clients:
                self._whole_train_graph(i)
        if conc:
            for i in local_subs:
                g = self._whole_train_graph(i)
                if g is None:
                    conc = False
                    break
                tgraphs[i] = g
        if conc:
            for i in local_subs:  # allocate on the main stream
                if i not in self._client_delta:
                    self._client_delta[i] = torch.empty_like(
                        self.global_flat)
            cur = torch.cuda.current_stream()
            costs = {}
            evs = {}
            th0 = time.perf_counter()
            for j, i in enumerate(local_subs):
                g, s = tgraphs[i], self._client_stream(j)
                s.wait_stream(cur)
                with torch.cuda.stream(s):
                    if self._phase_debug:
                        evs[i] = (torch.cuda.Event(enable_timing=True),
                                  torch.cuda.Event(enable_timing=True))
                        evs[i][0].record(s)
                    g.model.set_flat(self.global_flat)
                    costs[i] = g.run()
                    O.delta_extract_(self._client_delta[i],
                                     self.global_flat, g.model.flat.data,
                                     cfg.learning_rate)
                    if self._phase_debug:
                        evs[i][1].record(s)
            th1 = time.perf_counter()
            for j in range(len(local_subs)):
                cur.wait_stream(self._client_streams[j])
            if self._phase_debug:
                torch.cuda.synchronize(self.device)
                th2 = time.perf_counter()
                tt = " ".join(f"c{i}:{evs[i][0].elapsed_time(evs[i][1]):.1f}"
                              for i in local_subs)
                print(f"# [debug] epoch {epoch} train per-client ms: {tt} "
                      f"| host enqueue {1e3*(th1-th0):.2f} "
                      f"join {1e3*(th2-th1):.2f}", flush=True)
            # ONE host sync for every trainer's accumulated loss
            costl = torch.stack([costs[i] for i in local_subs]).tolist()
            for j, i in enumerate(local_subs):
                n = self.shards[i].n
                avg_cost = costl[j] / tgraphs[i].n_steps
                local_updates.append((self.origins[i],
                                      self._client_delta[i], n, avg_cost))
                samples_trained += n * cfg.local_epochs
        else:
            for i in local_subs:
                delta, n, cost = self._local_train(i)
                local_updates.append((self.origins[i], delta, n, cost))
                samples_trained += n * cfg.local_epochs
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        t1 = time.perf_counter()

        # ---- phase U: publish updates (one RCCL all-gather) -------------
        # each submission carries an HMAC tag binding (origin, epoch,
        # meta) to the origin's bootstrap key (chain/identity.py); every
        # replica verifies before feeding its ledger, so a forged origin
        # is rejected deterministically on all ranks (reference: one
        # ECDSA key per client, get_batch_accounts.sh)
        metas = [(o, epoch, n, c,
                  self.keys.sign("update", o, epoch, update_payload(n, c)))
                 for (o, delta, n, c) in local_updates]
        all_metas = self._gather("updates", self.t.all_gather_objects, metas)
        max_subs = max((len(m) for m in all_metas), default=0)
        P = self.global_flat.numel()
        rows = max(max_subs, 1)
        # cached, never zeroed: rows beyond a rank's own submission count
        # are never read back (unpack walks rank_metas, not the padding)
        if self._stack_buf is None or self._stack_buf.shape[0] < rows:
            self._stack_buf = torch.empty(rows, P, dtype=torch.float32,
                                          device=self.global_flat.device)
        stack = self._stack_buf[:rows]
        for j, (_, delta, _, _) in enumerate(local_updates):
            stack[j] = delta
        gathered = self._gather("update-deltas", self.t.all_gather_tensor,
                                stack)
        updates: List[Tuple[str, torch.Tensor]] = []
        n_bad_sig = 0
        for r, rank_metas in enumerate(all_metas):
            for j, (origin, ep, n, c, tag) in enumerate(rank_metas):
                if not self.keys.verify("update", origin, ep,
                                        update_payload(n, c), tag):
                    n_bad_sig += 1
                    continue
                code = led.upload_local_update(origin, b"", ep, n, c)
                if code == Admit.ACCEPTED:
                    updates.append((origin, gathered[r][j].to(self.device)))
        t2 = time.perf_counter()

        # ---- phase S: committee scoring ---------------------------------
        roles = led.roles()
        if cfg.self_scoring:
            local_scorers = list(self.shards.keys())
        else:
            local_scorers = [i for i in self.local_clients
                             if roles.get(self.origins[i]) == "comm"]
        packed = bool(local_scorers) and self._pack_candidates(updates)
        my_scores = []
        if self._concurrent and packed and self._round == 0:
            # pre-capture every local client's scoring graph (rotation:
            # each client eventually sits on the committee) so later
            # rounds never pay capture cost
            for i in self.local_clients:
                self._phase_scorer(i)
        # Concurrent path: every local scorer's whole-scoring graph (all
        # K candidates in one replay) runs on its own slot stream;
        # scores are read back once after the join.
        conc_s = self._concurrent and packed and len(local_scorers) > 1
        sgraphs = {}
        if conc_s:
            for i in local_scorers:
                g = self._phase_scorer(i)
                if g is None:
                    conc_s = False
                    break
                sgraphs[i] = g
        if conc_s:
            cur = torch.cuda.current_stream()
            res = {}
            evs = {}
            th0 = time.perf_counter()
            for j, i in enumerate(local_scorers):
                s = self._client_stream(j)
                s.wait_stream(cur)
                with torch.cuda.stream(s):
                    if self._phase_debug:
                        evs[i] = (torch.cuda.Event(enable_timing=True),
                                  torch.cuda.Event(enable_timing=True))
                        evs[i][0].record(s)
                    res[i] = sgraphs[i].run_inplace()
                    if self._phase_debug:
                        evs[i][1].record(s)
            th1 = time.perf_counter()
            for j in range(len(local_scorers)):
                cur.wait_stream(self._client_streams[j])
            if self._phase_debug:
                torch.cuda.synchronize(self.device)
                th2 = time.perf_counter()
                tt = " ".join(f"c{i}:{evs[i][0].elapsed_time(evs[i][1]):.1f}"
                              for i in local_scorers)
                print(f"# [debug] epoch {epoch} score per-client ms: {tt} "
                      f"| host enqueue {1e3*(th1-th0):.2f} "
                      f"join {1e3*(th2-th1):.2f}", flush=True)
            # ONE host sync for all scorers' [K] accuracy vectors
            alla = torch.stack([res[i] for i in local_scorers]).tolist()
            for j, i in enumerate(local_scorers):
                accs = alla[j]
                smap = {origin: accs[k]
                        for k, (origin, _) in enumerate(updates)}
                o = self.origins[i]
                my_scores.append((o, smap, self.keys.sign(
                    "scores", o, epoch, scores_payload(smap))))
        else:
            for i in local_scorers:
                o = self.origins[i]
                smap = self._score_candidates(i, updates, packed=packed)
                my_scores.append((o, smap, self.keys.sign(
                    "scores", o, epoch, scores_payload(smap))))
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        t3 = time.perf_counter()

        # ---- phase V: publish scores, decide ----------------------------
        all_scores = self._gather("scores", self.t.all_gather_objects,
                                  my_scores)
        decision = None
        for rank_scores in all_scores:
            for origin, smap, tag in rank_scores:
                if not self.keys.verify("scores", origin, epoch,
                                        scores_payload(smap), tag):
                    n_bad_sig += 1
                    continue
                d = led.upload_scores(origin, epoch, smap)
                if d is not None:
                    decision = d
        if decision is None:
            raise RuntimeError(
                f"round {epoch}: no aggregation decision "
                f"(scores={led.score_count}/{cfg.comm_count})")

        # ---- phase A: weighted FedAvg + commit --------------------------
        by_origin = dict(updates)
        sel = decision.selected
        K = len(sel)
        self.last_decision = decision
        if K > 0:
            if self._fedavg_buf is None or self._fedavg_buf.shape[0] < K:
                self._fedavg_buf = torch.empty(
                    K, P, dtype=torch.float32,
                    device=self.global_flat.device)
            deltas = self._fedavg_buf[:K]
            weights = torch.empty(K, dtype=torch.float32,
                                  device=self.global_flat.device)
            for k, (origin, w) in enumerate(sel):
                deltas[k] = by_origin[origin]
                weights[k] = float(w)
            avg = O.weighted_fedavg(deltas, weights)
            # global -= lr * avg  (reference .cpp:403-414)
            O.axpy_(self.global_flat, -cfg.learning_rate, avg)
        # K == 0 (no admitted updates survived): the model is unchanged
        # but the epoch still advances and the committee rotates/refills
        # — the round degrades instead of wedging every replica
        led.commit_aggregate(b"")
        if self.device.type == "cuda":
            torch.cuda.synchronize(self.device)
        t4 = time.perf_counter()

        acc = None
        if eval_global and self.test_shard is not None:
            acc = self.evaluate_global()
        self._round += 1
        stats = RoundStats(
            epoch=epoch, wall_s=t4 - t0, train_s=t1 - t0, gather_s=t2 - t1,
            score_s=t3 - t2, aggregate_s=t4 - t3,
            global_loss=led.global_loss, n_updates=len(updates),
            n_selected=K, samples_trained=samples_trained, test_acc=acc)
        self.metrics.log("round", stats,
                         selected=[o for o, _ in decision.selected],
                         bad_signatures=n_bad_sig)
        return stats

    # ------------------------------------------------------------------
    def evaluate_global(self) -> float:
        """Sponsor evaluation (reference run_sponsor, main.py:280-340)."""
        assert self.test_shard is not None
        self.model.set_flat(self.global_flat)
        return self.model.accuracy(self.test_shard.x, self.test_shard.y)

    def run(self, rounds: int, eval_every: int = 0) -> List[RoundStats]:
        out = []
        for r in range(rounds):
            if self.ledger.finished:  # max_epoch reached (main.py:251)
                break
            ev = eval_every > 0 and (r + 1) % eval_every == 0
            out.append(self.run_round(eval_global=ev))
        return out

    # ------------------------------------------------------------------
    # checkpoint / resume (chain persistence -> snapshot)
    def save(self, path: str) -> None:
        torch.save({
            "config": self.cfg.to_dict(),
            "ledger": self.ledger.snapshot(),
            "global_flat": self.global_flat.cpu(),
            "round": self._round,
        }, path)

    def load(self, path: str) -> None:
        ck = torch.load(path, map_location="cpu", weights_only=False)
        assert ck["config"]["model"] == self.cfg.model
        self.ledger.restore(ck["ledger"])
        # copy INTO the existing buffer: captured graphs bake its
        # device pointer (global_flat must never reallocate)
        with torch.no_grad():
            self.global_flat.copy_(ck["global_flat"].to(self.device))
        self._round = ck["round"]
