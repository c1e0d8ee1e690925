"""Engine-level GPU tests: long-run stability + checkpoint/resume
(the chain's persistence role) and accuracy progress on
the flagship config."""
import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = torch.device("cuda", 0)


def _engine(cfg):
    from bflc_amd.comm import Transport
    from bflc_amd.data import make_federated
    from bflc_amd.fl import FLEngine
    shards, test = make_federated(cfg)
    return FLEngine(cfg, Transport(device=DEV), shards, test)


def test_long_run_checkpoint_resume(tmp_path):
    from bflc_amd.config import FLConfig
    cfg = FLConfig.for_world(1, model="femnist_cnn", n_class=62,
                             samples_per_client=1024, batch_size=512,
                             eval_samples=1024, partition="dirichlet",
                             learning_rate=0.01)
    eng = _engine(cfg)
    eng.run(200)
    acc_mid = eng.evaluate_global()
    p = str(tmp_path / "ck.pt")
    eng.save(p)
    flat_at_save = eng.global_flat.clone()
    eng.run(25)

    # fresh engine, restore, must pick up bitwise where the save was
    eng2 = _engine(cfg)
    eng2.load(p)
    assert torch.equal(eng2.global_flat, flat_at_save)
    assert eng2.ledger.epoch == 200
    assert abs(eng2.evaluate_global() - acc_mid) < 1e-6
    eng2.run(25)
    assert eng2.ledger.epoch == 225
    # the flagship config LEARNS: synthetic FEMNIST at 200+ rounds is
    # well above the 62-class chance rate (~0.016; measured ~0.05 at
    # this reduced shard size, ~0.43 at the full bench config)
    assert acc_mid > 0.04


def test_graphed_step_matches_eager():
    """The hipGraph-captured train step (fl/graphs.py) replays exactly
    the eager kernel sequence: after the same rounds from the same seed
    the global models must agree."""
    from bflc_amd.config import FLConfig
    base = dict(model="femnist_cnn", n_class=62, samples_per_client=512,
                batch_size=256, eval_samples=512, partition="dirichlet",
                learning_rate=0.01)
    runs = {}
    for graphs in (True, False):
        cfg = FLConfig.for_world(1, use_graphs=graphs, **base)
        eng = _engine(cfg)
        eng.run(10)
        runs[graphs] = eng.global_flat.clone()
        if graphs:  # the graph path must actually have been taken
            assert any(g is not None
                       for g in eng._train_graphs.values()) \
                or len(eng._steppers) > 0, \
                "hipGraph capture fell back to eager on the GPU"
    assert torch.allclose(runs[True], runs[False], atol=1e-6, rtol=1e-5)


def test_graphed_adam_matches_eager():
    """Graphed Adam (device step counter + on-device bias corrections,
    csrc/hip/elementwise.hip adam_tick/adam_master_dev) must track the
    eager fused-Adam path (VERDICT round-1 item 10)."""
    from bflc_amd.config import FLConfig
    base = dict(model="femnist_cnn", n_class=62, samples_per_client=512,
                batch_size=256, eval_samples=512, partition="dirichlet",
                learning_rate=0.001, optimizer="adam")
    runs = {}
    for graphs in (True, False):
        cfg = FLConfig.for_world(1, use_graphs=graphs, **base)
        eng = _engine(cfg)
        eng.run(5)
        runs[graphs] = eng.global_flat.clone()
        if graphs:
            assert any(g is not None
                       for g in eng._train_graphs.values()) \
                or len(eng._steppers) > 0, \
                "graphed Adam fell back to eager on the GPU"
    # identical math modulo device-vs-host powf for the bias corrections
    assert torch.allclose(runs[True], runs[False], atol=1e-4, rtol=1e-4)


def test_adam_engine_matches_cpu_oracle():
    """Adam FL rounds on the GPU (bf16 compute, fused adam_master_
    kernels) against the engine's CPU fp32 Adam oracle: same seed, same
    data, same rounds. The compute dtypes differ (the GEMM family is
    bf16-only by design), so the assertion is on the round DELTA
    direction and magnitude — an Adam bug (swapped betas, wrong bias
    correction, stale m/v across clients) wrecks both; bf16 rounding
    does not (VERDICT round-1 item 10)."""
    from bflc_amd.comm import Transport
    from bflc_amd.config import FLConfig
    from bflc_amd.data import make_federated
    from bflc_amd.fl import FLEngine
    base = dict(model="mlp", n_features=16, n_class=4,
                samples_per_client=256, batch_size=128, eval_samples=512,
                optimizer="adam", learning_rate=0.001)
    deltas, accs = {}, {}
    for dev in (DEV, torch.device("cpu")):
        cfg = FLConfig.for_world(4, **base)
        shards, test = make_federated(cfg)
        eng = FLEngine(cfg, Transport(device=dev), shards, test)
        flat0 = eng.global_flat.clone()
        eng.run(3)
        deltas[dev.type] = (eng.global_flat - flat0).cpu()
        accs[dev.type] = eng.evaluate_global()
    cos = torch.nn.functional.cosine_similarity(
        deltas["cuda"], deltas["cpu"], dim=0)
    ratio = deltas["cuda"].norm() / deltas["cpu"].norm()
    assert float(cos) > 0.98, f"adam delta cosine {float(cos):.4f}"
    assert 0.9 < float(ratio) < 1.1, f"adam delta norm ratio {ratio:.3f}"
    assert abs(accs["cuda"] - accs["cpu"]) < 0.1


def test_byzantine_defense_on_gpu():
    """BASELINE config 4 on hardware: 2 of 8 clients label-flip; the
    committee-score defense must (a) never select an attacker update and
    (b) beat the attacked-FedAvg control (top-k widened to admit every
    update) on final accuracy (VERDICT round-1 item 7)."""
    from bflc_amd.config import FLConfig
    from bflc_amd.data import make_federated
    from bflc_amd.fl import FLEngine
    from bflc_amd.comm import Transport

    base = dict(client_num=8, comm_count=2, needed_update_count=6,
                byzantine_clients=2, model="mlp", n_features=32,
                n_class=8, samples_per_client=256, batch_size=128,
                eval_samples=1024, learning_rate=0.05)
    attackers = {"node_6", "node_7"}

    def run(agg):
        cfg = FLConfig(aggregate_count=agg, **base)
        shards, test = make_federated(cfg)
        assert shards[6].byzantine and shards[7].byzantine
        eng = FLEngine(cfg, Transport(device=DEV), shards, test)
        picked = []
        for _ in range(12):
            eng.run_round()
            picked += [o for o, _ in eng.last_decision.selected]
        return eng.evaluate_global(), picked

    defense_acc, defense_sel = run(agg=3)   # top-3 of 6: honest margin
    control_acc, control_sel = run(agg=6)   # everything aggregates
    assert not (set(defense_sel) & attackers), \
        f"attacker selected under defense: {set(defense_sel) & attackers}"
    assert set(control_sel) & attackers  # control really is attacked
    assert defense_acc > 0.9
    assert defense_acc > control_acc + 0.04, (defense_acc, control_acc)


def test_concurrent_streams_match_sequential(monkeypatch):
    """The per-client stream overlap (one model replica + one HIP
    stream per local client, concurrent whole-phase graph replays) must
    be a pure scheduling change: identical global model bits and
    identical committee decisions vs the sequential single-stream
    path."""
    from bflc_amd.config import FLConfig

    def run(streams: str):
        monkeypatch.setenv("BFLC_STREAMS", streams)
        cfg = FLConfig.for_world(8, model="femnist_cnn", n_class=62,
                                 samples_per_client=256, batch_size=128,
                                 eval_samples=256, partition="dirichlet")
        eng = _engine(cfg)
        sel = []
        for _ in range(6):
            eng.run_round()
            sel.append(tuple(o for o, _ in eng.last_decision.selected))
        return eng.global_flat.clone(), sel, eng

    flat_c, sel_c, eng_c = run("1")
    # the concurrent path really engaged: replicas + streams exist
    assert len(eng_c._client_streams) > 1, "no per-client streams made"
    assert len(eng_c._client_models) > 1, "no per-client replicas made"
    flat_s, sel_s, _ = run("0")
    assert sel_c == sel_s
    assert torch.equal(flat_c, flat_s)


def test_whole_phase_graphs_active_in_protocol_round():
    """The committee protocol round must actually run on the whole-phase
    graphs (one replay per client training pass, one per scorer) — not
    silently fall back to eager/per-batch paths."""
    from bflc_amd.config import FLConfig
    cfg = FLConfig.for_world(8, model="femnist_cnn", n_class=62,
                             samples_per_client=256, batch_size=128,
                             eval_samples=256, partition="dirichlet")
    eng = _engine(cfg)
    eng.run(2)
    assert any(g is not None for g in eng._train_graphs.values()), \
        "no whole-train graph captured"
    assert any(g is not None for g in eng._score_graphs.values()), \
        "no whole-scoring graph captured"
    # the packed candidate stack exists and matches the quota
    assert eng._cand_stack is not None
    assert eng._cand_stack.shape[0] == cfg.needed_update_count
