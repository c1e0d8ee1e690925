"""Multi-process distributed plumbing tests (gloo, CPU, world_size=2).

BASELINE config 1: "2-client FedAvg, 2-layer MLP on MNIST shards,
CPU/gloo world_size=2". Verifies that the distributed engine produces a
global model BITWISE identical on both ranks and identical to the
single-process run of the same config — the determinism property that
replaces PBFT replication.
"""
import json
import os
import subprocess
import sys

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import json, os, sys
import torch
sys.path.insert(0, {repo!r})
from bflc_amd.config import FLConfig
from bflc_amd.comm import Transport
from bflc_amd.data import make_federated
from bflc_amd.fl import FLEngine

cfg = FLConfig.for_world({world}, model="mlp", n_features=784, n_class=10,
                         samples_per_client=128, batch_size=32,
                         eval_samples=256, partition={partition!r},
                         byzantine_clients={byz})
shards, test = make_federated(cfg)
t = Transport(backend="gloo", device=torch.device("cpu"))
eng = FLEngine(cfg, t, shards, test)
eng.run({rounds})
out = {{
    "rank": t.rank,
    "epoch": eng.ledger.epoch,
    "roles": eng.ledger.roles(),
    "digest": torch.sum(eng.global_flat.double()).item(),
    "flat0": eng.global_flat[:8].tolist(),
    "acc": eng.evaluate_global(),
}}
with open(os.path.join({outdir!r}, f"rank{{t.rank}}.json"), "w") as f:
    json.dump(out, f)
t.barrier()
t.close()
"""


def run_world2(tmp_path, rounds=3, world=2, partition="iid", byz=0,
               port="29541"):
    script = tmp_path / "worker.py"
    script.write_text(WORKER.format(repo=REPO, rounds=rounds,
                                    outdir=str(tmp_path), world=world,
                                    partition=partition, byz=byz))
    procs = []
    for rank in range(world):
        env = dict(os.environ,
                   RANK=str(rank), WORLD_SIZE=str(world),
                   MASTER_ADDR="127.0.0.1", MASTER_PORT=port,
                   OMP_NUM_THREADS="2")
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    for p in procs:
        out, err = p.communicate(timeout=300)
        assert p.returncode == 0, err.decode()[-3000:]
    return [json.load(open(tmp_path / f"rank{r}.json"))
            for r in range(world)]


def test_world2_replicas_identical(tmp_path):
    r0, r1 = run_world2(tmp_path)
    assert r0["epoch"] == r1["epoch"] == 3
    assert r0["roles"] == r1["roles"]
    assert r0["digest"] == r1["digest"]
    assert r0["flat0"] == r1["flat0"]
    assert r0["acc"] == r1["acc"]


def test_world2_matches_single_process(tmp_path):
    """The distributed run must equal the world_size=1 run bit-for-bit
    (same clients, same feed order)."""
    from bflc_amd.config import FLConfig
    from bflc_amd.comm import Transport
    from bflc_amd.data import make_federated
    from bflc_amd.fl import FLEngine

    r0, _ = run_world2(tmp_path)

    cfg = FLConfig.for_world(2, model="mlp", n_features=784, n_class=10,
                             samples_per_client=128, batch_size=32,
                             eval_samples=256)
    shards, test = make_federated(cfg)
    eng = FLEngine(cfg, Transport(device=torch.device("cpu")), shards, test)
    eng.run(3)
    assert torch.sum(eng.global_flat.double()).item() == r0["digest"]
    assert eng.global_flat[:8].tolist() == r0["flat0"]


def test_world4_byzantine_replicas_identical(tmp_path):
    """4 ranks, non-IID shards, one label-flip attacker: the committee
    protocol must stay live and every replica must
    agree bitwise despite the hostile update."""
    res = run_world2(tmp_path, rounds=4, world=4, partition="dirichlet",
                     byz=1, port="29551")
    assert all(r["epoch"] == 4 for r in res)
    assert all(r["digest"] == res[0]["digest"] for r in res)
    assert all(r["roles"] == res[0]["roles"] for r in res)


# ---------------------------------------------------------------------------
# Straggler/crash tolerance (VERDICT round-1 item 5): the reference
# tolerated dead trainers by quota over-provisioning
# (CommitteePrecompiled.h:15 — any 10 of 16 suffice); the barrier-driven
# engine instead converts the hang into a diagnosable StragglerError
# with a rank-local checkpoint, bounded by the collective timeout.

STRAGGLER_WORKER = r"""
import os, sys, time
import torch
sys.path.insert(0, {repo!r})
from bflc_amd.config import FLConfig
from bflc_amd.comm import Transport
from bflc_amd.data import make_federated
from bflc_amd.fl import FLEngine, StragglerError

cfg = FLConfig.for_world(4, model="mlp", n_features=32, n_class=4,
                         samples_per_client=64, batch_size=32,
                         eval_samples=64)
shards, test = make_federated(cfg)
t = Transport(backend="gloo", device=torch.device("cpu"), timeout_s=15)
eng = FLEngine(cfg, t, shards, test)
eng.run_round()  # round 0 completes on all ranks
if t.rank == 3:
    os._exit(0)  # this rank dies before round 1
try:
    eng.run_round()
except StragglerError as e:
    print(f"STRAGGLER_CAUGHT rank={{t.rank}}: {{e}}", flush=True)
    sys.exit(7)
sys.exit(1)  # the round must NOT silently complete without rank 3
"""


def test_dead_rank_raises_straggler_error(tmp_path):
    import time as _time
    script = tmp_path / "straggler.py"
    script.write_text(STRAGGLER_WORKER.format(repo=REPO))
    ckpt = tmp_path / "abort_ckpt.pt"
    procs = []
    for rank in range(4):
        env = dict(os.environ, RANK=str(rank), WORLD_SIZE="4",
                   MASTER_ADDR="127.0.0.1", MASTER_PORT="29547",
                   OMP_NUM_THREADS="2",
                   BFLC_ABORT_CHECKPOINT=str(ckpt) + f".{rank}")
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    t0 = _time.time()
    outs = [p.communicate(timeout=180) for p in procs]
    elapsed = _time.time() - t0
    assert procs[3].returncode == 0  # the dead rank exited cleanly
    for r in range(3):
        out, err = outs[r]
        assert procs[r].returncode == 7, \
            (r, out.decode()[-500:], err.decode()[-2000:])
        msg = out.decode()
        assert "STRAGGLER_CAUGHT" in msg
        assert "dead or" in msg and "epoch 1" in msg
        # clean abort wrote the rank-local checkpoint
        assert os.path.exists(str(ckpt) + f".{r}")
    # diagnosable WITHIN the timeout budget, not an unbounded hang
    assert elapsed < 120


# ---------------------------------------------------------------------------
# Variable-length blob gather at ResNet-50 delta size (~102 MB): the
# padding/offset logic must round-trip big, unequal payloads (VERDICT
# round-1 item 8 — de-risk the first real 8-GPU run).

BIGBLOB_WORKER = r"""
import hashlib, os, sys
import torch
sys.path.insert(0, {repo!r})
from bflc_amd.comm import Transport

t = Transport(backend="gloo", device=torch.device("cpu"))
if t.rank == 0:
    blob = os.urandom(102_000_000)  # ~ResNet-50 flat fp32 delta
else:
    blob = os.urandom(1_234_567)    # very unequal: exercises padding
got = t.all_gather_blobs(blob)
assert len(got) == 2
assert got[t.rank] == blob
digests = [hashlib.sha256(b).hexdigest() for b in got]
lens = [len(b) for b in got]
print(f"RANK{{t.rank}} lens={{lens}} d0={{digests[0][:12]}} "
      f"d1={{digests[1][:12]}}", flush=True)
t.barrier()
t.close()
"""


def test_bigblob_gather_roundtrip(tmp_path):
    script = tmp_path / "bigblob.py"
    script.write_text(BIGBLOB_WORKER.format(repo=REPO))
    procs = []
    for rank in range(2):
        env = dict(os.environ, RANK=str(rank), WORLD_SIZE="2",
                   MASTER_ADDR="127.0.0.1", MASTER_PORT="29548",
                   OMP_NUM_THREADS="2")
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    lines = []
    for p in procs:
        out, err = p.communicate(timeout=300)
        assert p.returncode == 0, err.decode()[-3000:]
        lines.append(out.decode().strip())
    # both ranks saw the same bytes (identical digests, correct sizes)
    f0 = lines[0].split("lens=")[1]
    f1 = lines[1].split("lens=")[1]
    assert f0 == f1
    assert "[102000000, 1234567]" in lines[0]


# ---------------------------------------------------------------------------
# The exact bench protocol shape (8 clients / committee 4 / quota 4 /
# top-3) split over 4 ranks — the same client->rank split the driver's
# N=4 scaling point uses; replicas must stay bitwise identical.

BENCHSHAPE_WORKER = r"""
import json, os, sys
import torch
sys.path.insert(0, {repo!r})
from bflc_amd.config import FLConfig
from bflc_amd.comm import Transport
from bflc_amd.data import make_federated
from bflc_amd.fl import FLEngine

cfg = FLConfig.for_world(8, model="mlp", n_features=32, n_class=8,
                         samples_per_client=96, batch_size=32,
                         eval_samples=128, partition="dirichlet")
assert (cfg.client_num, cfg.comm_count, cfg.needed_update_count,
        cfg.aggregate_count) == (8, 4, 4, 3)
shards, test = make_federated(cfg)
t = Transport(backend="gloo", device=torch.device("cpu"))
eng = FLEngine(cfg, t, shards, test)
eng.run(4)
out = {{"rank": t.rank, "epoch": eng.ledger.epoch,
        "roles": eng.ledger.roles(),
        "digest": torch.sum(eng.global_flat.double()).item(),
        "local_clients": eng.local_clients}}
with open(os.path.join({outdir!r}, f"rank{{t.rank}}.json"), "w") as f:
    json.dump(out, f)
t.barrier()
t.close()
"""


def test_world4_bench_protocol_shape(tmp_path):
    script = tmp_path / "w4.py"
    script.write_text(BENCHSHAPE_WORKER.format(repo=REPO,
                                               outdir=str(tmp_path)))
    procs = []
    for rank in range(4):
        env = dict(os.environ, RANK=str(rank), WORLD_SIZE="4",
                   MASTER_ADDR="127.0.0.1", MASTER_PORT="29551",
                   OMP_NUM_THREADS="2")
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    for p in procs:
        out, err = p.communicate(timeout=300)
        assert p.returncode == 0, err.decode()[-3000:]
    rs = [json.load(open(tmp_path / f"rank{r}.json")) for r in range(4)]
    assert all(r["epoch"] == 4 for r in rs)
    assert all(r["digest"] == rs[0]["digest"] for r in rs)
    assert all(r["roles"] == rs[0]["roles"] for r in rs)
    # contiguous 2-client slices
    assert [r["local_clients"] for r in rs] == \
        [[0, 1], [2, 3], [4, 5], [6, 7]]
