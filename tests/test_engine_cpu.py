"""End-to-end FL engine tests on CPU (single process)."""
import pytest
import torch

from bflc_amd.config import FLConfig
from bflc_amd.comm import Transport
from bflc_amd.data import make_federated
from bflc_amd.fl import FLEngine, client_rank


def run_engine(cfg, rounds=5, eval_last=True):
    shards, test = make_federated(cfg)
    eng = FLEngine(cfg, Transport(), shards, test)
    stats = eng.run(rounds)
    acc = eng.evaluate_global() if eval_last else None
    return eng, stats, acc


class TestClientRank:
    def test_contiguous_cover(self):
        for n, w in [(20, 8), (8, 8), (20, 1), (7, 3), (1, 1)]:
            ranks = [client_rank(i, n, w) for i in range(n)]
            assert ranks == sorted(ranks)
            assert set(ranks) <= set(range(w))
            assert max(ranks) == min(w, n) - 1


class TestReferenceConfig:
    def test_20_client_logreg_learns(self):
        cfg = FLConfig()  # exact reference protocol constants
        eng, stats, acc = run_engine(cfg, rounds=8)
        assert stats[0].n_updates == 10
        assert stats[0].n_selected == 6
        assert stats[-1].epoch == 7
        # synthetic separable tabular data: linear model should be good
        assert acc > 0.85
        # loss decreases over rounds
        assert stats[-1].global_loss < stats[0].global_loss

    def test_rounds_advance_epochs(self):
        cfg = FLConfig(max_epoch=50)
        eng, stats, _ = run_engine(cfg, rounds=3, eval_last=False)
        assert [s.epoch for s in stats] == [0, 1, 2]
        assert eng.ledger.epoch == 3


class TestDeterminism:
    def test_two_runs_bitwise_identical(self):
        cfg = FLConfig()
        eng1, _, _ = run_engine(cfg, rounds=4, eval_last=False)
        eng2, _, _ = run_engine(cfg, rounds=4, eval_last=False)
        assert torch.equal(eng1.global_flat, eng2.global_flat)
        assert eng1.ledger.roles() == eng2.ledger.roles()


class TestWorldScaling:
    @pytest.mark.parametrize("n", [1, 2, 4, 8])
    def test_for_world_configs_run(self, n):
        cfg = FLConfig.for_world(n, model="mlp", n_features=32, n_class=10,
                                 samples_per_client=64, batch_size=32,
                                 eval_samples=128)
        eng, stats, acc = run_engine(cfg, rounds=2)
        assert stats[-1].epoch == 1
        assert eng.ledger.epoch == 2


class TestByzantine:
    def test_label_flip_attackers_excluded(self):
        # 8 nodes, 2 label-flip attackers (BASELINE config 4): committee
        # scoring must keep attacker updates out of the aggregate.
        cfg = FLConfig.for_world(8, model="mlp", n_features=16, n_class=4,
                                 samples_per_client=128, batch_size=32,
                                 byzantine_clients=2, local_epochs=2,
                                 learning_rate=0.05, eval_samples=256)
        # attackers are the LAST clients => node_6, node_7
        shards, test = make_federated(cfg)
        assert shards[6].byzantine and shards[7].byzantine
        eng = FLEngine(cfg, Transport(), shards, test)
        attackers = {"node_6", "node_7"}
        attacker_selected = honest_selected = 0
        for _ in range(6):
            eng.run_round()
            for origin, _ in eng.last_decision.selected:
                if origin in attackers:
                    attacker_selected += 1
                else:
                    honest_selected += 1
        # committee scoring is the defense: whenever an attacker's update
        # is among the candidates, its candidate scores low on honest
        # shards and is mostly excluded from the aggregate
        assert honest_selected > attacker_selected
        assert eng.evaluate_global() > 0.4  # above random (0.25)


class TestCheckpoint:
    def test_save_load_resume(self, tmp_path):
        cfg = FLConfig(max_epoch=50)
        shards, test = make_federated(cfg)
        eng = FLEngine(cfg, Transport(), shards, test)
        eng.run(3)
        p = str(tmp_path / "ck.pt")
        eng.save(p)
        flat_at_save = eng.global_flat.clone()
        eng.run(2)

        eng2 = FLEngine(cfg, Transport(), shards, test)
        eng2.load(p)
        assert torch.equal(eng2.global_flat, flat_at_save)
        assert eng2.ledger.epoch == 3
        eng2.run(2)
        assert torch.equal(eng2.global_flat, eng.global_flat)


class TestNonIID:
    def test_dirichlet_partition_runs(self):
        cfg = FLConfig.for_world(4, model="mlp", n_features=16, n_class=4,
                                 partition="dirichlet", dirichlet_alpha=0.3,
                                 samples_per_client=64, batch_size=16,
                                 eval_samples=128)
        eng, stats, acc = run_engine(cfg, rounds=2)
        assert stats[-1].epoch == 1


class TestDirichletBalance:
    def test_remainders_not_dumped_on_last_client(self):
        """Regression: per-class counts were floored with the summed
        remainder dumped on the LAST client — at many classes and few
        samples per class (ResNet-50 protocol shape: 1000 classes,
        ~2 samples each) client k-1 collected ~one sample per class
        (1375 of 2048) and its serial minibatch chain dominated every
        FL round. Cumulative rounding keeps every shard within the
        Dirichlet proportions."""
        import torch
        from bflc_amd.data.synthetic import partition_dirichlet
        gen = torch.Generator().manual_seed(0)
        n, k, n_class = 2048, 8, 1000
        x = torch.randn(n, 4)
        y = torch.randint(0, n_class, (n,), generator=gen)
        parts = partition_dirichlet(x, y, k, 0.3, n_class, gen)
        sizes = [p[0].shape[0] for p in parts]
        assert sum(sizes) == n
        # the buggy flooring gave the last client >half of everything;
        # proportional rounding keeps every shard near n/k at alpha-0.3
        # noise levels (the last client is NOT special)
        assert max(sizes) < n // 2, sizes
        assert sizes[-1] < 2 * n // k, sizes


class TestClientAssignment:
    def test_uneven_split_covers_all_clients(self):
        from bflc_amd.fl.engine import client_rank
        for n_clients in (1, 3, 7, 20):
            for world in (1, 2, 3, 8):
                owners = [client_rank(i, n_clients, world)
                          for i in range(n_clients)]
                # every client owned by exactly one valid rank,
                # contiguous ascending (the property the deterministic
                # admission order relies on)
                assert all(0 <= r < world for r in owners)
                assert owners == sorted(owners)
                if n_clients >= world:
                    assert set(owners) == set(range(world))


class TestMetricsJsonl:
    def test_per_round_records(self, tmp_path):
        import json

        import torch

        from bflc_amd.comm import Transport
        from bflc_amd.config import FLConfig
        from bflc_amd.data import make_federated
        from bflc_amd.fl import FLEngine

        path = str(tmp_path / "m.jsonl")
        cfg = FLConfig.for_world(1, model="logreg", n_features=8,
                                 n_class=2, samples_per_client=64,
                                 batch_size=32, eval_samples=64)
        eng = FLEngine(cfg, Transport(device=torch.device("cpu")),
                       *make_federated(cfg), metrics_path=path)
        eng.run(5, eval_every=2)
        recs = [json.loads(l) for l in open(path)]
        rounds = [r for r in recs if r["kind"] == "round"]
        assert len(rounds) == 5
        for r in rounds:
            for k in ("epoch", "wall_s", "train_s", "gather_s",
                      "score_s", "aggregate_s", "global_loss"):
                assert k in r, k
        # eval_every=2 attaches test_acc to those rounds
        assert sum(1 for r in rounds if r.get("test_acc") is not None) >= 2


def test_reference_default_constants():
    """The reference's protocol constants survive as the default profile
    (CommitteePrecompiled.h:6-19, main.py:52-88)."""
    from bflc_amd.config import FLConfig
    c = FLConfig.reference_defaults()
    assert c.client_num == 20
    assert c.comm_count == 4
    assert c.needed_update_count == 10
    assert c.aggregate_count == 6
    assert c.learning_rate == 1e-3
    assert c.batch_size == 100
    assert c.n_features == 5 and c.n_class == 2


def test_random_config_protocol_soak():
    """Randomized protocol-liveness sweep: any well-formed config must
    run N rounds with epochs advancing, a full committee after every
    rotation, and two runs bitwise identical (the PBFT-replacement
    property)."""
    try:
        from hypothesis import given, settings, strategies as st
    except ImportError:
        import pytest
        pytest.skip("hypothesis not installed")
    import torch

    from bflc_amd.comm import Transport
    from bflc_amd.config import FLConfig
    from bflc_amd.data import make_federated
    from bflc_amd.fl import FLEngine

    @settings(max_examples=12, deadline=None)
    @given(data=st.data())
    def check(data):
        n = data.draw(st.integers(min_value=2, max_value=8))
        comm = data.draw(st.integers(min_value=1,
                                     max_value=max(1, n // 2)))
        trainers = n - comm
        quota = data.draw(st.integers(min_value=comm, max_value=trainers))
        agg = data.draw(st.integers(min_value=1, max_value=quota))
        byz = data.draw(st.integers(min_value=0, max_value=n - 1))
        part = data.draw(st.sampled_from(["iid", "dirichlet"]))
        cfg = FLConfig(client_num=n, comm_count=comm,
                       needed_update_count=quota, aggregate_count=agg,
                       byzantine_clients=byz, partition=part,
                       model="logreg", n_features=6, n_class=2,
                       samples_per_client=32, batch_size=16,
                       eval_samples=32)
        runs = []
        for _ in range(2):
            eng = FLEngine(cfg, Transport(device=torch.device("cpu")),
                           *make_federated(cfg))
            eng.run(8)
            assert eng.ledger.epoch == 8
            roles = eng.ledger.roles()
            assert sum(1 for r in roles.values() if r == "comm") == comm
            runs.append(eng.global_flat.clone())
        assert torch.equal(runs[0], runs[1])

    check()


class TestDegradedRound:
    def test_all_forged_updates_degrade_not_wedge(self):
        """If EVERY update submission fails HMAC verification (hostile
        transport / mis-bound keys), zero candidates are admitted —
        the round must still complete: empty decision, unchanged
        model, epoch advances, committee rotates, and the next clean
        round recovers. (The reference's chain would simply not reach
        its quota and poll forever; the barrier engine degrades
        instead.)"""
        cfg = FLConfig.for_world(4, model="logreg", n_features=8,
                                 n_class=2, samples_per_client=64,
                                 batch_size=32, eval_samples=64)
        shards, test = make_federated(cfg)
        eng = FLEngine(cfg, Transport(), shards, test)
        flat0 = eng.global_flat.clone()
        from bflc_amd.chain.identity import KeyTable
        orig = eng.keys
        wrong = KeyTable(eng.origins, seed=999)  # attacker's keys

        class SplitKeys:
            """Sign updates with the WRONG key table (an attacker
            without the bootstrap seed); verify with the true one."""
            def sign(self, kind, o, e, p):
                return (wrong if kind == "update" else orig).sign(
                    kind, o, e, p)

            def verify(self, *a):
                return orig.verify(*a)

        eng.keys = SplitKeys()
        st = eng.run_round()
        assert st.n_updates == 0 and st.n_selected == 0
        assert torch.equal(eng.global_flat, flat0)  # model unchanged
        assert eng.ledger.epoch == 1  # epoch still advanced

        eng.keys = orig  # clean round recovers
        st2 = eng.run_round()
        assert st2.n_updates > 0 and st2.n_selected > 0
        assert not torch.equal(eng.global_flat, flat0)


class TestMaxEpoch:
    """max_epoch is enforced end-to-end (reference main.py:65,251-252:
    clients exit once epoch > MAX_EPOCH) — VERDICT round-1 missing #6."""

    def test_run_stops_at_max_epoch(self):
        cfg = FLConfig.for_world(4, model="logreg", samples_per_client=64,
                                 batch_size=32, eval_samples=64,
                                 max_epoch=2)
        eng, stats, _ = run_engine(cfg, rounds=10, eval_last=False)
        # epochs 0,1,2 ran; epoch 3 > max_epoch stops the loop
        assert len(stats) == 3
        assert eng.ledger.epoch == 3
        assert eng.ledger.finished

    def test_run_round_raises_after_finish(self):
        cfg = FLConfig.for_world(2, model="logreg", samples_per_client=64,
                                 batch_size=32, eval_samples=64,
                                 max_epoch=0)
        eng, stats, _ = run_engine(cfg, rounds=5, eval_last=False)
        assert len(stats) == 1
        with pytest.raises(RuntimeError, match="max_epoch"):
            eng.run_round()
