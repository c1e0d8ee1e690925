"""Property tests for the deterministic committee ledger.

Mirrors the reference contract's state machine semantics
(CommitteePrecompiled.cpp:132-311, Aggregate .cpp:349-455): epoch
guards, duplicate rejection, update quota, committee-only scoring,
median -> top-k -> rotation, and checkpoint/restore.
"""
import pytest

from bflc_amd._ledger import Admit, CommitteeLedger, LedgerConfig, median_ref


def make_ledger(n=20, comm=4, needed=10, agg=6):
    lc = LedgerConfig()
    lc.client_num, lc.comm_count = n, comm
    lc.needed_update_count, lc.aggregate_count = needed, agg
    return CommitteeLedger(lc)


def register_all(led, n=20):
    for i in range(n):
        led.register_node(f"node_{i}")


class TestRegistration:
    def test_epoch_starts_uninitialized(self):
        led = make_ledger()
        assert led.epoch == -999  # reference .cpp:322

    def test_epoch_zero_after_full_registration(self):
        led = make_ledger()
        register_all(led)
        assert led.epoch == 0

    def test_first_registrants_become_committee(self):
        led = make_ledger()
        register_all(led)
        assert led.committee() == [f"node_{i}" for i in range(4)]
        assert len(led.trainers()) == 16

    def test_duplicate_registration_ignored(self):
        led = make_ledger()
        register_all(led)
        assert led.register_node("node_0") is False
        assert led.epoch == 0

    def test_unregistered_query_state_defaults_trainer(self):
        led = make_ledger()
        role, ep = led.query_state("ghost")
        assert role == "trainer" and ep == -999  # reference .cpp:196-199


class TestUpdateAdmission:
    def test_reject_before_start(self):
        led = make_ledger()
        assert led.upload_local_update("node_0", b"", -999, 1, 0.0) == \
            Admit.NOT_STARTED

    def test_stale_epoch_rejected(self):
        led = make_ledger()
        register_all(led)
        assert led.upload_local_update("node_5", b"", 1, 10, 0.0) == \
            Admit.STALE_EPOCH  # reference .cpp:225-226

    def test_duplicate_rejected(self):
        led = make_ledger()
        register_all(led)
        assert led.upload_local_update("node_5", b"a", 0, 10, 0.0) == \
            Admit.ACCEPTED
        assert led.upload_local_update("node_5", b"b", 0, 10, 0.0) == \
            Admit.DUPLICATE  # reference .cpp:232-233

    def test_quota(self):
        led = make_ledger()
        register_all(led)
        for i in range(10):
            assert led.upload_local_update(f"node_{4+i}", b"", 0, 10, 0.0) \
                == Admit.ACCEPTED
        assert led.upload_local_update("node_15", b"", 0, 10, 0.0) == \
            Admit.QUOTA_FULL  # reference .cpp:239-244

    def test_query_all_updates_empty_until_quota(self):
        led = make_ledger()
        register_all(led)
        led.upload_local_update("node_5", b"x", 0, 10, 0.0)
        assert led.query_all_updates() == []  # reference .cpp:304-307
        for i in range(9):
            led.upload_local_update(f"node_{6+i}", b"", 0, 10, 0.0)
        ups = led.query_all_updates()
        assert len(ups) == 10
        assert ups[0] == ("node_5", b"x")  # acceptance order


class TestScoring:
    def _filled(self):
        led = make_ledger()
        register_all(led)
        for i in range(10):
            led.upload_local_update(f"node_{4+i}", b"", 0, 100 + i, 0.5)
        return led

    def test_trainer_scores_rejected(self):
        led = self._filled()
        assert led.upload_scores("node_10", 0, {"node_4": 1.0}) is None
        assert led.score_count == 0  # reference .cpp:272-275

    def test_stale_epoch_scores_rejected(self):
        led = self._filled()
        assert led.upload_scores("node_0", 1, {"node_4": 1.0}) is None
        assert led.score_count == 0  # reference .cpp:266-269

    def test_overwrite_does_not_double_count(self):
        led = self._filled()
        led.upload_scores("node_0", 0, {"node_4": 1.0})
        led.upload_scores("node_0", 0, {"node_4": 0.9})
        assert led.score_count == 1  # divergence: reference .cpp:279-289

    def test_aggregation_on_last_committee_score(self):
        led = self._filled()
        trainers = [f"node_{4+i}" for i in range(10)]
        for c in range(3):
            scores = {t: 0.5 + 0.01 * i for i, t in enumerate(trainers)}
            assert led.upload_scores(f"node_{c}", 0, scores) is None
        scores = {t: 0.5 + 0.01 * i for i, t in enumerate(trainers)}
        dec = led.upload_scores("node_3", 0, scores)
        assert dec is not None  # reference .cpp:296-297
        # top 6 by median score: node_13..node_8 in descending order
        assert [s for s, _ in dec.selected] == \
            [f"node_{13 - i}" for i in range(6)]
        assert dec.total_weight == sum(100 + i for i in range(9, 3, -1))
        # next committee = top 4
        assert dec.next_committee == [f"node_{13 - i}" for i in range(4)]

    def test_commit_rotates_roles_and_clears(self):
        led = self._filled()
        trainers = [f"node_{4+i}" for i in range(10)]
        for c in range(4):
            scores = {t: float(i) for i, t in enumerate(trainers)}
            dec = led.upload_scores(f"node_{c}", 0, scores)
        led.commit_aggregate(b"G1")
        assert led.epoch == 1
        assert led.update_count == 0 and led.score_count == 0
        assert led.query_all_updates() == []
        comm = led.committee()
        assert comm == sorted([f"node_{13 - i}" for i in range(4)])
        # old committee all back to trainer (reference .cpp:443-455)
        for c in range(4):
            assert led.query_state(f"node_{c}")[0] == "trainer"

    def test_median_tiebreak_deterministic(self):
        led = self._filled()
        trainers = [f"node_{4+i}" for i in range(10)]
        for c in range(4):
            dec = led.upload_scores(f"node_{c}", 0,
                                    {t: 0.5 for t in trainers})
        # all tied -> lexicographic id order
        assert [s for s, _ in dec.selected] == sorted(trainers)[:6]


class TestMedian:
    def test_even(self):
        assert median_ref([1.0, 2.0, 3.0, 4.0]) == 2.5

    def test_odd(self):
        assert median_ref([3.0, 1.0, 2.0]) == 2.0

    def test_reference_getmid_semantics(self):
        # GetMid (.cpp:81-115): even n averages the two middles
        assert median_ref([0.9, 0.1, 0.5, 0.7]) == pytest.approx(0.6)

    def test_single(self):
        assert median_ref([0.42]) == pytest.approx(0.42)


class TestDeterminism:
    def test_same_feed_same_state(self):
        def run():
            led = make_ledger()
            register_all(led)
            for i in range(10):
                led.upload_local_update(f"node_{4+i}", b"", 0, 50 + i, 0.1)
            trainers = [f"node_{4+i}" for i in range(10)]
            for c in range(4):
                dec = led.upload_scores(
                    f"node_{c}", 0,
                    {t: (hash(t + str(c)) % 100) / 100 for t in trainers})
            led.commit_aggregate(b"G")
            return led.snapshot()

        a, b = run(), run()
        assert a["roles"] == b["roles"]
        assert a["epoch"] == b["epoch"]


class TestSnapshot:
    def test_roundtrip(self):
        led = make_ledger()
        register_all(led)
        led.set_global_model(b"MODEL")
        led.upload_local_update("node_7", b"U", 0, 33, 0.25)
        snap = led.snapshot()

        led2 = make_ledger()
        led2.restore(snap)
        assert led2.epoch == 0
        assert led2.update_count == 1
        assert led2.query_global_model() == (b"MODEL", 0)
        assert led2.roles() == led.roles()
        assert led2.update_meta("node_7") == (33, 0.25)
        # duplicate still rejected after restore
        assert led2.upload_local_update("node_7", b"U", 0, 33, 0.25) == \
            Admit.DUPLICATE


class TestSelfScoringWorld1:
    def test_single_node_world(self):
        led = make_ledger(n=1, comm=1, needed=1, agg=1)
        led.register_node("node_0")
        assert led.epoch == 0
        assert led.query_state("node_0")[0] == "comm"
        assert led.upload_local_update("node_0", b"", 0, 10, 0.5) == \
            Admit.ACCEPTED
        dec = led.upload_scores("node_0", 0, {"node_0": 0.8})
        assert dec is not None and dec.selected == [("node_0", 10)]
        led.commit_aggregate(b"")
        assert led.epoch == 1


class TestProtocolTrace:
    def test_reference_log_lines(self):
        """BFLC_LEDGER_TRACE=1 reprints the reference's OUTPUT-gated
        markers (CommitteePrecompiled.h:4, .cpp:255-257,291-293,
        422-425) from the replica."""
        import subprocess
        import sys
        code = (
            "import os; os.environ['BFLC_LEDGER_TRACE']='1';"
            "from bflc_amd._ledger import CommitteeLedger;"
            "from bflc_amd.config import FLConfig;"
            "cfg=FLConfig(client_num=4,comm_count=1,needed_update_count=2,"
            "aggregate_count=1);"
            "led=CommitteeLedger(cfg.ledger_config());"
            "[led.register_node(f'node_{i}') for i in range(4)];"
            "led.set_global_model(b'');"
            "led.upload_local_update('node_1',b'',0,10,0.5);"
            "led.upload_local_update('node_2',b'',0,10,0.4);"
            "led.upload_scores('node_0',0,{'node_1':0.9,'node_2':0.8});"
            "led.commit_aggregate(b'')")
        r = subprocess.run([sys.executable, "-c", code],
                           capture_output=True, text=True, timeout=120)
        assert r.returncode == 0, r.stderr
        assert "the update of local model is collected!" in r.stderr
        assert "scores has been uploaded" in r.stderr
        assert "epoch , global loss :" in r.stderr


class TestHostileInput:
    """Round-2 hardening: no input sequence may throw from upload/
    decide/commit (VERDICT weak #3), and max_epoch is enforced
    (reference main.py:251-252 exits when epoch > MAX_EPOCH)."""

    def _run_round(self, led, n_up=10, extra_scores=None):
        epoch = led.epoch
        ups = [f"node_{i}" for i in range(4, 4 + n_up)]
        for o in ups:
            led.upload_local_update(o, b"", epoch, 10, 0.5)
        decision = None
        for o in led.committee():
            smap = {u: 0.5 for u in ups}
            if extra_scores:
                smap.update(extra_scores)
            d = led.upload_scores(o, epoch, smap)
            if d is not None:
                decision = d
        return decision

    def test_unknown_id_scores_are_dropped(self):
        led = make_ledger()
        register_all(led)
        led.set_global_model(b"")
        d = self._run_round(led, extra_scores={"ghost": 9.9, "": 1.0})
        assert d is not None
        selected = [o for o, _ in d.selected]
        assert "ghost" not in selected and "" not in selected
        assert "ghost" not in d.median_scores
        led.commit_aggregate(b"")  # must not throw
        assert led.epoch == 1

    def test_empty_score_maps_never_throw(self):
        led = make_ledger()
        register_all(led)
        led.set_global_model(b"")
        epoch = led.epoch
        for o in [f"node_{i}" for i in range(4, 14)]:
            led.upload_local_update(o, b"", epoch, 10, 0.5)
        d = None
        for o in led.committee():
            d = led.upload_scores(o, epoch, {}) or d
        # all-empty maps: every admitted trainer gets median 0.0 and the
        # top-k is still well-defined (id-ascending tiebreak)
        assert d is not None
        assert len(d.selected) == 6
        assert all(v == 0.0 for v in d.median_scores.values())
        led.commit_aggregate(b"")
        assert led.epoch == 1

    def test_partial_score_map_defaults_missing_to_zero(self):
        led = make_ledger(n=6, comm=2, needed=3, agg=3)
        register_all(led, 6)
        led.set_global_model(b"")
        for o in ("node_2", "node_3", "node_4"):
            led.upload_local_update(o, b"", 0, 10, 0.5)
        led.upload_scores("node_0", 0, {"node_2": 1.0})  # omits 3, 4
        d = led.upload_scores("node_1", 0,
                              {"node_2": 1.0, "node_3": 0.8, "node_4": 0.6})
        assert d is not None
        # median over [1.0, 1.0] / [0.0, 0.8] / [0.0, 0.6]
        assert d.median_scores["node_2"] == pytest.approx(1.0)
        assert d.median_scores["node_3"] == pytest.approx(0.4)
        assert d.median_scores["node_4"] == pytest.approx(0.3)

    def test_rotated_out_member_scores_rejected(self):
        led = make_ledger()
        register_all(led)
        led.set_global_model(b"")
        old_comm = led.committee()
        d = self._run_round(led)
        led.commit_aggregate(b"")
        rotated_out = [o for o in old_comm if o not in led.committee()]
        if not rotated_out:  # rotation kept everyone (scored top-4)
            pytest.skip("no member rotated out in this round")
        # a stale-committee member's scores are silently refused
        assert led.upload_scores(rotated_out[0], led.epoch,
                                 {"node_5": 1.0}) is None
        assert led.score_count == 0

    def test_max_epoch_closes_the_ledger(self):
        led = make_ledger(n=4, comm=1, needed=3, agg=2)
        # drive a tiny ledger past max_epoch
        lc = LedgerConfig()
        lc.client_num, lc.comm_count = 4, 1
        lc.needed_update_count, lc.aggregate_count = 3, 2
        lc.max_epoch = 1
        led = CommitteeLedger(lc)
        register_all(led, 4)
        led.set_global_model(b"")
        for _ in range(2):  # epochs 0 and 1 run normally
            epoch = led.epoch
            ups = [o for o in led.trainers()][:3]
            for o in ups:
                assert led.upload_local_update(o, b"", epoch, 10, .5) \
                    == Admit.ACCEPTED
            d = None
            for o in led.committee():
                d = led.upload_scores(o, epoch, {u: .5 for u in ups}) or d
            assert d is not None
            led.commit_aggregate(b"")
        assert led.epoch == 2 and led.finished
        # epoch 2 > max_epoch 1: everything is refused, nothing throws
        assert led.upload_local_update("node_3", b"", 2, 10, .5) \
            == Admit.FINISHED
        assert led.upload_scores(led.committee()[0], 2, {}) is None
        assert led.score_count == 0
