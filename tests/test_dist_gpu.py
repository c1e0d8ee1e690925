"""Multi-process engine test ON GPU TENSORS (gloo transport, world 2,
both ranks sharing cuda:0 — a 1-GPU box can't run 2 RCCL ranks, but the
full engine phase sequence over CUDA tensors + cross-process gathers is
exactly what the first real 8-GPU run will execute, with only the
backend swapped; VERDICT round-1 item 8)."""
import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import json, os, sys
import torch
sys.path.insert(0, {repo!r})
from bflc_amd.config import FLConfig
from bflc_amd.comm import Transport
from bflc_amd.data import make_federated
from bflc_amd.fl import FLEngine

cfg = FLConfig.for_world(8, model="femnist_cnn", n_class=62,
                         samples_per_client=256, batch_size=128,
                         eval_samples=256, partition="dirichlet")
shards, test = make_federated(cfg)
# both ranks on the one visible GPU; transport stays gloo (CPU staging
# inside Transport for the gathers), data plane tensors live on cuda:0
t = Transport(backend="gloo", device=torch.device("cuda", 0))
eng = FLEngine(cfg, t, shards, test)
eng.run(3)
out = {{
    "rank": t.rank,
    "epoch": eng.ledger.epoch,
    "roles": eng.ledger.roles(),
    "digest": torch.sum(eng.global_flat.double()).item(),
    "acc": eng.evaluate_global(),
    "device": str(eng.global_flat.device),
}}
with open(os.path.join({outdir!r}, f"rank{{t.rank}}.json"), "w") as f:
    json.dump(out, f)
t.barrier()
t.close()
"""


def test_world2_gpu_tensors_gloo(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER.format(repo=REPO, outdir=str(tmp_path)))
    procs = []
    for rank in range(2):
        env = dict(os.environ, RANK=str(rank), WORLD_SIZE="2",
                   MASTER_ADDR="127.0.0.1", MASTER_PORT="29561")
        procs.append(subprocess.Popen(
            [sys.executable, str(script)], env=env,
            stdout=subprocess.PIPE, stderr=subprocess.PIPE))
    for p in procs:
        out, err = p.communicate(timeout=600)
        assert p.returncode == 0, err.decode()[-3000:]
    r0, r1 = [json.load(open(tmp_path / f"rank{r}.json")) for r in (0, 1)]
    assert r0["epoch"] == r1["epoch"] == 3
    assert r0["roles"] == r1["roles"]
    assert r0["digest"] == r1["digest"]  # bitwise-identical replicas
    assert r0["acc"] == r1["acc"]
    assert r0["device"].startswith("cuda")
