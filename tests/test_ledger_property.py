"""Property-based fuzz of the C++ committee ledger against a pure-Python
model of the reference contract semantics
(reference CommitteePrecompiled.cpp:168-455): random interleavings of
registrations, uploads (stale/duplicate/over-quota) and scores must
produce identical admission decisions, epoch advances and role
rotations."""
import pytest

try:
    from hypothesis import given, settings, strategies as st
except ImportError:  # pragma: no cover
    pytest.skip("hypothesis not installed", allow_module_level=True)

import numpy as np

from bflc_amd._ledger import Admit, CommitteeLedger
from bflc_amd.config import FLConfig


class PyLedgerModel:
    """Reference semantics in plain Python (the oracle)."""

    def __init__(self, cfg):
        self.cfg = cfg
        self.registered = []
        self.epoch = -999
        self.roles = {}
        self.updates = {}   # origin -> (n, cost)
        self.scores = {}    # origin -> {trainer: score}

    def register(self, origin):
        if self.epoch != -999 or origin in self.registered:
            return
        self.registered.append(origin)
        if len(self.registered) == self.cfg.client_num:
            self.epoch = 0
            # our documented divergence: committee = FIRST registrants
            for i, o in enumerate(self.registered):
                self.roles[o] = ("comm" if i < self.cfg.comm_count
                                 else "trainer")
            if self.cfg.client_num == 1:
                self.roles[self.registered[0]] = "trainer"

    def upload(self, origin, epoch, n, cost):
        if self.epoch < 0 or epoch != self.epoch:
            return "stale"
        if origin in self.updates:
            return "duplicate"
        if len(self.updates) >= self.cfg.needed_update_count:
            return "quota"
        self.updates[origin] = (n, cost)
        return "accepted"

    def upload_scores(self, origin, epoch, smap):
        if self.epoch < 0 or epoch != self.epoch:
            return None
        role = self.roles.get(origin, "trainer")
        scorer_ok = role == "comm" or self.cfg.client_num == 1
        if not scorer_ok:
            return None
        self.scores[origin] = dict(smap)
        need = (1 if self.cfg.client_num == 1 else self.cfg.comm_count)
        if len(self.scores) < need:
            return None
        # aggregate: median per trainer (in fp32 — the ledger keeps
        # the reference's float rounding, csrc/ledger.cpp:81),
        # top aggregate_count
        meds = {}
        for tr in self.updates:
            vals = sorted(np.float32(s.get(tr, 0.0))
                          for s in self.scores.values())
            k = len(vals)
            if k % 2 == 1:
                m = vals[k // 2]
            else:  # reference GetMid averages the two middles
                m = np.float32(0.5) * (vals[k // 2 - 1] + vals[k // 2])
            meds[tr] = float(m)
        ranked = sorted(meds.items(), key=lambda kv: (-kv[1], kv[0]))
        sel = [o for o, _ in ranked[: self.cfg.aggregate_count]]
        return sel

    def commit(self):
        # role rotation: all comm -> trainer; top comm_count scored
        # trainers -> next committee
        meds = {}
        for tr in self.updates:
            vals = sorted(np.float32(s.get(tr, 0.0))
                          for s in self.scores.values())
            k = len(vals)
            m = vals[k // 2] if k % 2 else \
                np.float32(0.5) * (vals[k // 2 - 1] + vals[k // 2])
            meds[tr] = float(m)
        ranked = sorted(meds.items(), key=lambda kv: (-kv[1], kv[0]))
        new_comm = [o for o, _ in ranked[: self.cfg.comm_count]]
        # refill short committees in registration order (ledger
        # anti-deadlock divergence, csrc/ledger.cpp commit_aggregate)
        for o in self.registered:
            if len(new_comm) >= self.cfg.comm_count:
                break
            if o not in new_comm:
                new_comm.append(o)
        if self.cfg.client_num > 1:
            for o in self.roles:
                self.roles[o] = ("comm" if o in new_comm else "trainer")
        self.updates.clear()
        self.scores.clear()
        self.epoch += 1


@settings(max_examples=60, deadline=None)
@given(data=st.data())
def test_ledger_matches_python_model(data):
    n_clients = data.draw(st.integers(min_value=2, max_value=8))
    comm = data.draw(st.integers(min_value=1,
                                 max_value=min(4, n_clients // 2)))
    trainers = n_clients - comm
    needed = data.draw(st.integers(min_value=comm, max_value=trainers))
    agg = data.draw(st.integers(min_value=1, max_value=needed))
    cfg = FLConfig(client_num=n_clients, comm_count=comm,
                   needed_update_count=needed, aggregate_count=agg)
    led = CommitteeLedger(cfg.ledger_config())
    model = PyLedgerModel(cfg)
    origins = [f"node_{i}" for i in range(n_clients)]
    for o in origins:
        led.register_node(o)
        model.register(o)
    led.set_global_model(b"")
    assert led.epoch == model.epoch == 0

    for _ in range(data.draw(st.integers(min_value=1, max_value=4))):
        # trainers upload in a random order, sometimes stale/duplicate.
        # The first upload each round is always fresh: with ZERO
        # admitted updates the ledger refuses to aggregate (the
        # reference would divide by zero at .cpp:397 in that state),
        # and the engine guarantees quota admissions per round.
        order = data.draw(st.permutations(origins))
        for idx, o in enumerate(order):
            ep = model.epoch + (0 if idx == 0 else
                                data.draw(st.sampled_from([0, 0, 0, -1, 1])))
            n = data.draw(st.integers(min_value=1, max_value=500))
            code = led.upload_local_update(o, b"", ep, n, 0.5)
            ref = model.upload(o, ep, n, 0.5)
            assert (code == Admit.ACCEPTED) == (ref == "accepted"), \
                (o, ep, code, ref)
        # committee members score every admitted update
        decision = None
        for o in origins:
            if model.roles.get(o) != "comm":
                continue
            smap = {tr: data.draw(st.floats(min_value=0, max_value=1,
                                            allow_nan=False, width=32))
                    for tr in model.updates}
            # hostile input: sometimes key scores by ids with no admitted
            # update — the ledger must drop them at admission and the
            # round must proceed identically (VERDICT weak #3)
            if data.draw(st.booleans()):
                smap[data.draw(st.sampled_from(
                    ["ghost", "", "node_999", origins[0] + "x"]))] = \
                    data.draw(st.floats(min_value=0, max_value=10,
                                        allow_nan=False, width=32))
            d = led.upload_scores(o, model.epoch, smap)
            ref_sel = model.upload_scores(o, model.epoch, smap)
            if ref_sel is not None:
                decision = d
                assert d is not None
                assert [x for x, _ in d.selected] == ref_sel
            else:
                assert d is None
        assert decision is not None
        led.commit_aggregate(b"")
        model.commit()
        assert led.epoch == model.epoch
        assert led.roles() == model.roles
